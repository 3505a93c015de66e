"""C++ router core (_router_core): hashing parity, prefix index, queues."""
import numpy as np
import pytest
import xxhash

from llm_d_inference_scheduler_amd import _router_core as rc


class TestXXH64:
    @pytest.mark.parametrize("data,seed", [
        (b"", 0), (b"a", 0), (b"abc", 1), (b"x" * 31, 7), (b"y" * 32, 7),
        (b"z" * 33, 99), (bytes(range(256)) * 5, 2**63),
    ])
    def test_parity_with_reference_lib(self, data, seed):
        assert rc.xxh64(data, seed) == xxhash.xxh64(data, seed=seed).intdigest()


class TestHashTokens:
    def test_chain_definition(self):
        seed0 = rc.model_seed("m", "s")
        toks = np.arange(64, dtype=np.int32)
        h = rc.hash_tokens(toks, 16, 256, seed0)
        assert len(h) == 4
        prev = seed0
        for b in range(4):
            c = xxhash.xxh64(toks[b * 16:(b + 1) * 16].tobytes(),
                             seed=seed0).intdigest()
            prev = xxhash.xxh64(
                np.array([c, prev], dtype=np.uint64).tobytes()).intdigest()
            assert h[b] == prev

    def test_partial_block_dropped(self):
        h = rc.hash_tokens(np.arange(17, dtype=np.int32), 16, 256, 1)
        assert len(h) == 1

    def test_max_blocks(self):
        h = rc.hash_tokens(np.arange(160, dtype=np.int32), 16, 4, 1)
        assert len(h) == 4

    def test_prefix_property(self):
        """Shared prefixes share leading hashes; divergence changes the tail."""
        seed0 = rc.model_seed("llama", "")
        a = np.arange(64, dtype=np.int32)
        b = a.copy(); b[48] += 1
        ha = rc.hash_tokens(a, 16, 256, seed0)
        hb = rc.hash_tokens(b, 16, 256, seed0)
        assert list(ha[:3]) == list(hb[:3])
        assert ha[3] != hb[3]

    def test_model_seed_differs(self):
        toks = np.arange(16, dtype=np.int32)
        h1 = rc.hash_tokens(toks, 16, 256, rc.model_seed("a", ""))
        h2 = rc.hash_tokens(toks, 16, 256, rc.model_seed("b", ""))
        assert h1[0] != h2[0]


class TestPrefixIndex:
    def test_match_longest(self):
        idx = rc.PrefixIndex(1000)
        h = rc.hash_tokens(np.arange(160, dtype=np.int32), 16, 256, 3)
        idx.add(0, h[:10])
        idx.add(1, h[:5])
        idx.add(5, h[:1])
        counts = idx.match_longest(h, 6)
        assert list(counts) == [10, 5, 0, 0, 0, 1]

    def test_consecutive_requirement(self):
        idx = rc.PrefixIndex(1000)
        h = rc.hash_tokens(np.arange(80, dtype=np.int32), 16, 256, 3)
        # endpoint holds blocks 0,1 and 3,4 but not 2 -> match stops at 2
        idx.add(0, np.concatenate([h[:2], h[3:]]))
        counts = idx.match_longest(h, 1)
        assert counts[0] == 2

    def test_lru_eviction_and_size(self):
        idx = rc.PrefixIndex(4)
        h = rc.hash_tokens(np.arange(160, dtype=np.int32), 16, 256, 3)
        evicted = idx.add(0, h)          # 10 blocks into capacity 4
        assert evicted == 6
        assert idx.endpoint_size(0) == 4
        assert idx.size() == 4
        # remaining entries are the most recently added (tail of the chain)
        counts = idx.match_longest(h, 1)
        assert counts[0] == 0  # head evicted -> no leading match

    def test_remove_endpoint(self):
        idx = rc.PrefixIndex(100)
        h = rc.hash_tokens(np.arange(32, dtype=np.int32), 16, 256, 3)
        idx.add(0, h); idx.add(1, h)
        idx.remove_endpoint(0)
        counts = idx.match_longest(h, 2)
        assert list(counts) == [0, 2]
        assert idx.size() == 2  # endpoint 1 still holds them

    def test_touch_refreshes_lru(self):
        idx = rc.PrefixIndex(2)
        h = rc.hash_tokens(np.arange(48, dtype=np.int32), 16, 256, 3)
        idx.add(0, h[:2])
        idx.add(0, h[:1])   # touch block 0 -> block 1 becomes LRU victim
        idx.add(0, h[2:3])  # evicts block 1
        counts = idx.match_longest(h, 1)
        assert counts[0] == 1


class TestQueues:
    def test_listqueue_fifo(self):
        q = rc.ListQueue()
        for i in range(5):
            q.push(i + 1, 0.0, 10 * (i + 1))
        assert len(q) == 5 and q.bytes == 150
        assert q.peek() == 1 and q.peek_tail() == 5
        assert q.pop()[0] == 1
        assert q.remove(3) == 30
        assert [q.pop()[0] for _ in range(3)] == [2, 4, 5]
        assert q.pop() is None and q.bytes == 0

    def test_maxminheap(self):
        q = rc.MaxMinHeap()
        import random
        rng = random.Random(0)
        keys = {i: rng.random() for i in range(1, 101)}
        for i, k in keys.items():
            q.push(i, k, 1)
        assert q.peek() == min(keys, key=keys.get)
        assert q.peek_max() == max(keys, key=keys.get)
        out = []
        while len(q):
            out.append(q.pop()[0])
        assert out == sorted(keys, key=keys.get)

    def test_heap_remove_arbitrary(self):
        q = rc.MaxMinHeap()
        for i in range(1, 11):
            q.push(i, float(i), i)
        assert q.remove(5) == 5
        assert q.remove(5) is None
        out = [q.pop()[0] for _ in range(len(q))]
        assert out == [1, 2, 3, 4, 6, 7, 8, 9, 10]


class TestProfileRunner:
    def _run(self, scorers, picker=0, **kw):
        E = kw.get("E", 4)
        pr = rc.ProfileRunner(kw.get("seed", 1))
        return pr.run(
            kw.get("roles", np.full(E, 1, np.uint8)),
            kw.get("queue", np.zeros(E, np.float32)),
            kw.get("running", np.zeros(E, np.float32)),
            kw.get("kv", np.zeros(E, np.float32)),
            kw.get("tokens", np.zeros(E, np.float32)),
            kw.get("active", np.zeros(E, np.float32)),
            kw.get("role_filter", 0), kw.get("mask"), scorers,
            kw.get("match"), kw.get("total", 0), kw.get("extra"),
            picker, kw.get("max_endpoints", 1))

    def test_queue_scorer_minmax(self):
        picks, scores = self._run([(0, 1.0, 0.0, 0.0)],
                                  queue=np.array([0, 10, 5, 10], np.float32))
        assert picks[0] == 0
        assert scores[0] == 1.0 and scores[1] == 0.0 and abs(scores[2] - .5) < 1e-6

    def test_kv_and_prefix(self):
        picks, scores = self._run(
            [(1, 1.0, 0.0, 0.0), (2, 2.0, 0.0, 0.0)],
            kv=np.array([0.5, 0.2, 0.9, 0.0], np.float32),
            match=np.array([0, 4, 0, 0], np.int32), total=8)
        # ep1: (1-0.2) + 2*0.5 = 1.8 -> best
        assert picks[0] == 1
        assert abs(scores[1] - 1.8) < 1e-5

    def test_role_filter(self):
        roles = np.array([1, 1, 2, 2], np.uint8)  # 2 decode, 2 prefill
        picks, scores = self._run([(0, 1.0, 0.0, 0.0)], roles=roles, role_filter=2,
                                  queue=np.array([0, 0, 9, 1], np.float32))
        assert picks[0] == 3
        assert scores[0] == -1.0 and scores[1] == -1.0  # filtered

    def test_candidate_mask(self):
        picks, _ = self._run([(0, 1.0, 0.0, 0.0)],
                             mask=np.array([0, 0, 1, 0], np.uint8))
        assert picks[0] == 2

    def test_weighted_random_distribution(self):
        E = 3
        counts = np.zeros(E)
        pr = rc.ProfileRunner(7)
        extra = np.array([0.7, 0.2, 0.1], np.float32)
        for _ in range(3000):
            picks, _ = pr.run(np.full(E, 1, np.uint8),
                              np.zeros(E, np.float32), np.zeros(E, np.float32),
                              np.zeros(E, np.float32), np.zeros(E, np.float32),
                              np.zeros(E, np.float32),
                              0, None, [], None, 0, extra, 2, 1)
            counts[picks[0]] += 1
        freq = counts / counts.sum()
        assert freq[0] > freq[1] > freq[2]
        assert abs(freq[0] - 0.7) < 0.06

    def test_empty_after_filter(self):
        picks, scores = self._run([(0, 1.0, 0.0, 0.0)], role_filter=4)  # no encode
        assert len(picks) == 0


def test_sanitizer_clean():
    """ASan+UBSan pass over the C++ core (SURVEY 5.2 `-race` analog).
    Compiles csrc/router/test/sanitize_main.cpp and runs its randomized
    workloads; any heap error / UB aborts with nonzero exit."""
    import os
    import shutil
    import subprocess
    if shutil.which("g++") is None:
        import pytest
        pytest.skip("no g++")
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(["bash", os.path.join(repo, "tools",
                                             "sanitize_check.sh")],
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, (r.stdout + r.stderr)[-2000:]
    assert "sanitize: OK" in r.stdout
