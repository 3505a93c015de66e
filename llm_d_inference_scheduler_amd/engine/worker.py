"""Per-GPU worker engine: continuous batching over the paged KV pool.

This is the MI355X replacement for the reference's external vLLM model-server
pods (the router only steers them). One EngineWorker owns
one GPU: chunked prefill + batched decode (gfx950 paged-attention kernel),
role-aware behavior (decode | prefill | encode | combinations), and the
vLLM-compatible metrics snapshot the datalayer collectors scrape.

Decode hot loop design (MI355X-first): all per-step state — block tables,
sequence lengths, last sampled token ids — is device-resident; a step is
assembled and launched with no host→device traffic except on block-boundary
crossings and batch-membership changes, and sampled tokens are read back one
step late through a pinned buffer + event (the host never blocks the GPU on
a same-step sync). Finish decisions are made from host-side token *counts*,
so step N+1 launches before step N's token values have landed.
"""
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np
import torch

from ..datalayer.endpoint import Metrics, Role
from ..models.configs import ModelConfig
from ..models.llama import ForwardBatch, LlamaRunner
from ..utils.logging import get_logger
from .kvcache import BLOCK_SIZE, BlockManager, KVPool, block_hashes

log = get_logger("engine.worker")


@dataclass
class EngineRequest:
    request_id: str
    prompt_tokens: List[int]
    max_tokens: int = 16
    temperature: float = 0.0
    is_embedding: bool = False
    prefill_only: bool = False      # disagg: stop after prefill + 1st token
    cached_tokens: int = 0          # prefix-cache reuse (engine-measured)
    priority: int = 0               # InferenceObjective priority (preemption)
    stop_token_ids: Optional[List[int]] = None  # finish_reason "stop"
    # multimodal: embeddings for the first len(prefix_embeds) prompt rows
    # (prompt_tokens must carry placeholder ids for those positions)
    prefix_embeds: "Optional[torch.Tensor]" = None
    # runtime state
    computed: int = 0               # prompt tokens already prefilled
    generated: List[int] = field(default_factory=list)
    inflight: int = 0               # sampled on device, not yet collected
    slo_ok: bool = True             # TTFT met the SLO (goodput accounting)
    arrival_t: float = 0.0
    first_token_t: float = 0.0
    block_hashes: Optional[object] = None   # np.uint64 per full prompt block
    registered_blocks: int = 0              # hashes published so far
    _prompt_np: Optional[object] = field(default=None, repr=False)

    def __post_init__(self):
        if not self.arrival_t:
            self.arrival_t = time.time()

    @property
    def prompt_len(self) -> int:
        return len(self.prompt_tokens)

    @property
    def token_count(self) -> int:
        """Generated + in-flight tokens (host-known without a sync)."""
        return len(self.generated) + self.inflight


@dataclass
class RequestOutput:
    request_id: str
    new_tokens: List[int]
    finished: bool = False
    finish_reason: str = "length"   # length | stop
    kind: str = "decode"            # decode | prefill_done | embedding
    # prefill_done payload (disagg hand-off):
    kv_blocks: Optional[List[int]] = None
    seq_len: int = 0
    first_token: int = 0
    # embedding payload:
    embedding: Optional[torch.Tensor] = None
    all_tokens: Optional[List[int]] = None  # full generation, on finish
    # usage on finish:
    prompt_tokens: int = 0
    completion_tokens: int = 0
    cached_tokens: int = 0
    ttft_ms: Optional[float] = None
    tpot_ms: Optional[float] = None
    e2e_ms: Optional[float] = None
    error: str = ""                 # terminal engine-side failure


class DecodeState:
    """Device-resident decode batch state (grows on demand)."""

    def __init__(self, capacity: int, max_blocks: int, device: torch.device):
        self.device = device
        self.max_blocks = max_blocks
        self.capacity = 0
        self.bt = None
        self.seq_lens = None
        self.last_tok = None
        self.temps = None
        self.free_rows: List[int] = []
        self.row_of: Dict[str, int] = {}
        self._active_ids: List[str] = []
        self._rows_t: Optional[torch.Tensor] = None
        self._grow(capacity)

    def _grow(self, new_cap: int) -> None:
        old = self.capacity
        bt = torch.zeros((new_cap, self.max_blocks), dtype=torch.int32,
                         device=self.device)
        sl = torch.zeros(new_cap, dtype=torch.int32, device=self.device)
        lt = torch.zeros(new_cap, dtype=torch.int64, device=self.device)
        tp = torch.zeros(new_cap, dtype=torch.float32, device=self.device)
        if old:
            bt[:old] = self.bt
            sl[:old] = self.seq_lens
            lt[:old] = self.last_tok
            tp[:old] = self.temps
        self.bt, self.seq_lens, self.last_tok, self.temps = bt, sl, lt, tp
        self.free_rows.extend(range(new_cap - 1, old - 1, -1))
        self.capacity = new_cap

    def take_row(self) -> int:
        if not self.free_rows:
            self._grow(max(8, self.capacity * 2))
        return self.free_rows.pop()

    def join(self, req: EngineRequest, table: List[int], seq_len: int,
             last_token: int) -> None:
        row = self.take_row()
        self.row_of[req.request_id] = row
        # any membership mutation invalidates the cached row-index tensor:
        # an identical id list can recur with different row assignments
        # after a leave/rejoin (chunked-decode continuations)
        self._active_ids = []
        self._rows_t = None
        if len(table) > self.max_blocks:
            raise RuntimeError("sequence exceeds decode-state max_blocks")
        self.bt[row, :len(table)] = torch.tensor(
            table, dtype=torch.int32).to(self.device, non_blocking=True)
        self.seq_lens[row] = seq_len
        self.last_tok[row] = last_token
        self.temps[row] = req.temperature

    def leave(self, request_id: str) -> None:
        row = self.row_of.pop(request_id, None)
        if row is not None:
            self.free_rows.append(row)
            self._active_ids = []
            self._rows_t = None

    def rows_for(self, ids: List[str]) -> torch.Tensor:
        if ids != self._active_ids:
            self._rows_t = torch.tensor(
                [self.row_of[i] for i in ids], dtype=torch.int64,
                device=self.device)
            self._active_ids = list(ids)
        return self._rows_t


class EngineWorker:
    def __init__(self, config: ModelConfig, device,
                 role: Role = Role.DECODE,
                 kv_blocks: Optional[int] = None,
                 kv_budget_bytes: int = 8 << 30,
                 dtype: torch.dtype = torch.bfloat16,
                 prefill_chunk_tokens: int = 8192,
                 max_decode_batch: int = 256,
                 max_model_len: int = 8192,
                 ttft_slo_ms: float = None,
                 prefix_caching: bool = True,
                 kv_cache_dtype: str = "auto",
                 prefill_min_tokens: int = 4096,
                 prefill_max_delay_ms: float = 60.0,
                 ipc_pool: bool = False,
                 overlap_streams: Optional[bool] = None,
                 seed: int = 0):
        self.cfg = config
        self.device = torch.device(device)
        self._is_cuda_dev = lambda: self.device.type == "cuda"
        self.role = role
        self.dtype = dtype
        self.prefill_chunk_tokens = prefill_chunk_tokens
        self.prefill_min_tokens = prefill_min_tokens
        self.prefill_max_delay_ms = prefill_max_delay_ms
        self.max_decode_batch = max_decode_batch
        self.max_model_len = max_model_len
        self.ttft_slo_ms = ttft_slo_ms
        cache_dtype = {"auto": dtype, "bf16": torch.bfloat16,
                       "fp8": torch.float8_e4m3fn}[kv_cache_dtype]
        if cache_dtype == torch.float8_e4m3fn and \
                self.device.type != "cuda":
            cache_dtype = dtype     # fp8 path needs the gfx950 kernels
        if kv_blocks is None:
            kv_blocks = KVPool.blocks_for_budget(
                config, kv_budget_bytes,
                dtype_bytes=cache_dtype.itemsize)
        self.pool = KVPool(config, kv_blocks, self.device, dtype,
                           cache_dtype=cache_dtype,
                           ipc_alloc=ipc_pool and self._is_cuda_dev())
        self.mgr = BlockManager(kv_blocks, self.pool.block_size,
                                prefix_caching=prefix_caching)
        self.model = LlamaRunner(config, self.device, dtype, seed=seed)
        self.waiting: List[EngineRequest] = []
        self.running: List[EngineRequest] = []
        self._by_id: Dict[str, EngineRequest] = {}
        # rows must hold prompt + generation; add_request clamps
        # prompt + max_tokens <= max_model_len, so this bound is exact
        max_blocks = max_model_len // self.pool.block_size + 2
        self.dstate = DecodeState(min(64, max_decode_batch), max_blocks,
                                  self.device)
        import os as _os
        self._cuda = self.device.type == "cuda"
        # Prefill/decode stream overlap measured +11.8% at mixed
        # prefill/decode shapes (profiles/r02_notes.md) but wedged two
        # long open-loop runs non-deterministically (sweep r30 mono, r20
        # fc — suspect a shared-resource race between concurrent GEMMs on
        # the two streams). Tri-state: explicit True/False from the
        # constructor wins (tests force it on); the None default is
        # OFF unless LLMD_ENABLE_OVERLAP=1, until the wedge is
        # root-caused.
        if overlap_streams is None:
            self._overlap = _os.environ.get("LLMD_ENABLE_OVERLAP",
                                            "0") == "1"
        else:
            self._overlap = bool(overlap_streams)
        self._prefill_stream = None   # lazy; see step() overlap
        self._pin = None
        self._pending = None   # (reqs, event|None, n) — one-step readback lag
        self.steps = 0
        self._rejects: List[RequestOutput] = []
        self._carry: List[RequestOutput] = []   # outputs surfaced early
        self.total_generated = 0
        self.total_generated_slo = 0
        self.total_prefilled = 0

    # ------------------------------------------------------------------
    def _reject(self, req: EngineRequest, error: str) -> None:
        self._rejects.append(RequestOutput(
            request_id=req.request_id, new_tokens=[], finished=True,
            finish_reason="error", prompt_tokens=len(req.prompt_tokens),
            error=error))

    def add_request(self, req: EngineRequest) -> None:
        if not req.prompt_tokens:
            # an empty prompt would be silently dropped in _prefill_pass
            # (remaining <= 0) and the client would hang until timeout
            self._reject(req, "empty_prompt: prompt tokenized to 0 tokens")
            return
        if len(req.prompt_tokens) >= self.max_model_len:
            # reject, never truncate: a silently-truncated prompt produces
            # a generation the caller cannot interpret (vLLM analog:
            # context_length_exceeded)
            self._reject(
                req, f"context_length_exceeded: prompt "
                     f"{len(req.prompt_tokens)} >= max_model_len "
                     f"{self.max_model_len}")
            return
        if any(t >= self.cfg.vocab_size or t < 0
               for t in req.prompt_tokens):
            # defensive clamp: a mismatched tokenizer must not crash the
            # engine loop (ids fold into the vocab deterministically)
            req.prompt_tokens = [t % self.cfg.vocab_size
                                 for t in req.prompt_tokens]
        capacity = self.pool.num_blocks * self.pool.block_size
        if len(req.prompt_tokens) >= capacity:
            # the prompt alone can never fit: reject instead of the
            # infinite preempt/recompute loop the engine-lifecycle fuzz
            # found (vLLM analog: scheduler watermark rejection)
            self._reject(
                req, f"kv_capacity_exceeded: prompt "
                     f"{len(req.prompt_tokens)} >= pool {capacity} tokens")
            return
        # generation ceiling: both the KV pool AND the model context window
        # (positions past max_model_len walk off the RoPE table and past
        # DecodeState's block-table width)
        ceiling = min(capacity, self.max_model_len)
        if len(req.prompt_tokens) + req.max_tokens > ceiling:
            clamped = ceiling - len(req.prompt_tokens)
            log.warning("max_tokens clamped to capacity",
                        req=req.request_id, requested=req.max_tokens,
                        clamped=clamped)
            req.max_tokens = clamped
        self.waiting.append(req)
        self._by_id[req.request_id] = req

    def admit_transferred(self, req: EngineRequest, local_blocks: List[int],
                          seq_len: int, first_token: int) -> None:
        """Adopt a prefilled sequence whose KV arrived over xGMI. The
        adopted full blocks are registered in the prefix cache: a
        transferred prefix is as reusable as a locally-computed one."""
        self.mgr.adopt(req.request_id, local_blocks, seq_len)
        if req.block_hashes is None:
            req.block_hashes = block_hashes(req.prompt_tokens,
                                            self.pool.block_size)
        full = min(len(local_blocks), req.prompt_len // self.pool.block_size)
        for b in range(full):
            self.mgr.register_block(req.request_id, b,
                                    int(req.block_hashes[b]))
        req.computed = req.prompt_len
        req.generated = [first_token]
        if not req.first_token_t:
            req.first_token_t = time.time()
        if self.ttft_slo_ms is not None and req.arrival_t and \
                (req.first_token_t - req.arrival_t) * 1e3 > self.ttft_slo_ms:
            req.slo_ok = False
        self._by_id[req.request_id] = req
        self.running.append(req)
        self.dstate.join(req, local_blocks, seq_len, first_token)

    def abort(self, request_id: str) -> None:
        req = self._by_id.pop(request_id, None)
        if req is None:
            return
        if req in self.waiting:
            self.waiting.remove(req)
        if req in self.running:
            self.running.remove(req)
        self.dstate.leave(request_id)
        self.mgr.free(request_id)

    # ------------------------------------------------------------------
    def metrics_snapshot(self) -> Metrics:
        return Metrics(
            waiting_queue_size=len(self.waiting),
            running_requests_size=len(self.running),
            kv_cache_usage=self.mgr.usage,
            cache_block_size=self.pool.block_size,
            cache_num_blocks=self.pool.num_blocks,
        )

    @property
    def has_work(self) -> bool:
        return bool(self.waiting or self.running or self._pending
                    or self._rejects)

    # ------------------------------------------------------------------
    def step(self) -> List[RequestOutput]:
        """One engine iteration: a chunked-prefill pass (if any waiting) and
        one decode pass over the running batch. Decode token values surface
        one step late (pipelined readback)."""
        outputs: List[RequestOutput] = []
        if self._rejects:
            outputs.extend(self._rejects)
            self._rejects = []
        if self._carry:
            outputs.extend(self._carry)
            self._carry = []
        do_prefill = bool(self.waiting) and self._should_prefill()
        if self._cuda and self._overlap and do_prefill and self.running:
            # Overlap the two passes: decode (HBM-bound paged attention +
            # skinny GEMMs) launches first on the default stream; prefill
            # (MFMA-bound flash attention + fat GEMMs) runs concurrently
            # on its own stream. Safe because the passes touch DISJOINT
            # sequences and KV blocks (freed blocks re-enter the pool only
            # after their final reader's event synchronized, and cached
            # prefix blocks are immutable), and weights are read-only.
            # No ps.wait_stream(default) on entry — that would serialize
            # the two passes and defeat the overlap.
            outputs.extend(self._decode_pass())
            if self._prefill_stream is None:
                self._prefill_stream = torch.cuda.Stream(self.device)
            ps = self._prefill_stream
            with torch.cuda.stream(ps):
                outputs.extend(self._prefill_pass())
            # _prefill_pass host-syncs ps for first-token sampling, but
            # dstate.join() enqueues block-table copies on ps AFTER that
            # sync point: the next step's decode (default stream) must
            # observe them
            torch.cuda.current_stream(self.device).wait_stream(ps)
        else:
            if do_prefill:
                outputs.extend(self._prefill_pass())
            if self.running:
                outputs.extend(self._decode_pass())
            elif self._pending is not None:
                outputs.extend(self._collect_pending())
        self.steps += 1
        return outputs

    def _should_prefill(self) -> bool:
        """Batch prefill work: a small per-step chunk (one trickling
        arrival) launches a half-empty grid every layer; accumulating
        arrivals to `prefill_min_tokens` fills the chip and amortizes the
        pass, bounded by `prefill_max_delay_ms` of added TTFT."""
        if not self.running:
            return True                    # decode idle (or prefill role)
        if self.waiting[0].computed > 0:
            return True                    # finish a split chunk promptly
        pending = sum(r.prompt_len - r.computed for r in self.waiting)
        if pending >= self.prefill_min_tokens:
            return True
        oldest = self.waiting[0].arrival_t
        return oldest > 0 and \
            (time.time() - oldest) * 1e3 >= self.prefill_max_delay_ms

    # ---- prefill ----
    def _admit_prompt(self, req: EngineRequest) -> None:
        """First touch: match + reuse cached prefix blocks (engine APC)."""
        if req.block_hashes is None:
            req.block_hashes = block_hashes(req.prompt_tokens,
                                            self.pool.block_size)
        matched = self.mgr.allocate_prompt(req.request_id, req.block_hashes,
                                           req.prompt_len)
        req.computed = matched
        req.cached_tokens = matched
        req.registered_blocks = matched // self.pool.block_size
        self.mgr.set_seq_len(req.request_id, matched)

    def _prefill_pass(self) -> List[RequestOutput]:
        budget = self.prefill_chunk_tokens
        selected: List[tuple] = []
        # priority-ordered admission (stable within a priority class):
        # a preempted victim must not re-grab blocks freed FOR a
        # higher-priority waiter
        for req in sorted(self.waiting, key=lambda r: -r.priority):
            if budget <= 0:
                break
            if req.request_id not in self.mgr.tables:
                self._admit_prompt(req)
            remaining = req.prompt_len - req.computed
            if remaining <= 0:
                self.waiting.remove(req)
                continue
            chunk = min(remaining, budget)
            if not self.mgr.can_allocate(req.computed + chunk,
                                         req.request_id) and not selected:
                # cannot even fit one: a higher-priority arrival preempts
                # the lowest-priority running request NOW instead of
                # stalling until full KV exhaustion (InferenceObjective
                # priority semantics; round-1 notes refinement)
                self._maybe_preempt_for(req)
                break
            if not self.mgr.allocate(req.request_id, req.computed + chunk):
                break
            selected.append((req, chunk))
            budget -= chunk
        if not selected:
            return []

        ids_np, pos_np, slots_np = [], [], []
        seq_starts, ctx_lens, tables, logit_rows = [0], [], [], []
        embed_rows, embed_vals = [], []
        finishing: List[EngineRequest] = []
        bs = self.pool.block_size
        metas = []                      # (seq_start, chunk, prior) per seq
        for req, chunk in selected:
            start, end = req.computed, req.computed + chunk
            metas.append((seq_starts[-1], chunk, start))
            if req.prefix_embeds is not None and start < req.prefix_embeds.shape[0]:
                e_end = min(end, req.prefix_embeds.shape[0])
                base_row = seq_starts[-1] + 0
                embed_rows.extend(range(base_row, base_row + (e_end - start)))
                embed_vals.append(req.prefix_embeds[start:e_end])
            # vectorized assembly (a per-token python loop here was ~ms per
            # 4k-token pass and left the GPU idle at steady state)
            if req._prompt_np is None or len(req._prompt_np) != req.prompt_len:
                req._prompt_np = np.asarray(req.prompt_tokens, dtype=np.int64)
            ids_np.append(req._prompt_np[start:end])
            p = np.arange(start, end, dtype=np.int64)
            pos_np.append(p)
            table_np = np.asarray(self.mgr.tables[req.request_id],
                                  dtype=np.int64)
            slots_np.append(table_np[p // bs] * bs + p % bs)
            self.mgr.set_seq_len(req.request_id, end)
            ctx_lens.append(end)
            tables.append(self.mgr.tables[req.request_id])
            seq_starts.append(seq_starts[-1] + chunk)
            if end == req.prompt_len:
                logit_rows.append(seq_starts[-1] - 1)
                finishing.append(req)
            req.computed = end
            self.total_prefilled += chunk
            # publish content hashes of now-complete blocks (prefill only;
            # shared cached blocks are immutable by construction)
            full = end // bs
            for b in range(req.registered_blocks, full):
                self.mgr.register_block(req.request_id, b,
                                        int(req.block_hashes[b]))
            req.registered_blocks = max(req.registered_blocks, full)
        input_ids = np.concatenate(ids_np)
        positions = np.concatenate(pos_np)
        slots = np.concatenate(slots_np)

        qpg = self.cfg.num_heads // self.cfg.num_kv_heads
        flash_ok = (self._cuda and self.cfg.head_dim == 128
                    and qpg in (1, 2, 4, 8)
                    and self.dtype == torch.bfloat16)
        prefill_bt = prefill_meta = prefill_tiles = None
        bt_list = None
        if flash_ok:
            maxb = max(len(t) for t in tables)
            bt_np = np.zeros((len(tables), maxb), dtype=np.int32)
            for i, t in enumerate(tables):
                bt_np[i, :len(t)] = t
            prefill_bt = torch.from_numpy(bt_np).to(self.device,
                                                    non_blocking=True)
            prefill_meta = torch.tensor(metas, dtype=torch.int32).to(
                self.device, non_blocking=True)
            tl = [(i, v0) for i, (_, chunk, _) in enumerate(metas)
                  for v0 in range(0, chunk * qpg, 128)]
            prefill_tiles = torch.tensor(tl, dtype=torch.int32).to(
                self.device, non_blocking=True)
        else:
            bt_list = [torch.tensor(t, dtype=torch.int32,
                                    device=self.device) for t in tables]
        batch = ForwardBatch(
            input_ids=torch.from_numpy(input_ids).to(self.device,
                                                     non_blocking=True),
            positions=torch.from_numpy(
                positions.astype(np.int32)).to(self.device,
                                               non_blocking=True),
            slot_mapping=torch.from_numpy(slots).to(self.device,
                                                    non_blocking=True),
            is_decode=False, seq_starts=seq_starts, ctx_lens=ctx_lens,
            prefill_block_tables=bt_list,
            prefill_bt=prefill_bt, prefill_meta=prefill_meta,
            prefill_tiles=prefill_tiles,
            logit_rows=torch.tensor(logit_rows, dtype=torch.int64,
                                    device=self.device)
            if logit_rows else torch.zeros(0, dtype=torch.int64,
                                           device=self.device),
            embed_rows=torch.tensor(embed_rows, dtype=torch.int64,
                                    device=self.device)
            if embed_rows else None,
            embed_values=torch.cat(embed_vals).to(self.device)
            if embed_vals else None)
        embedding_reqs = [r for r in finishing if r.is_embedding]
        hidden_or_logits = self.model.forward(
            batch, self.pool.tensor, embeddings_out=bool(embedding_reqs))

        outputs: List[RequestOutput] = []
        now = time.time()
        if embedding_reqs:
            # embeddings: mean-pool each finished sequence's chunk rows
            for i, (req, chunk) in enumerate(selected):
                if req not in embedding_reqs:
                    continue
                rows = hidden_or_logits[seq_starts[i]:seq_starts[i + 1]]
                emb = rows.float().mean(dim=0)
                outputs.append(RequestOutput(
                    request_id=req.request_id, new_tokens=[], finished=True,
                    kind="embedding", embedding=emb.cpu(),
                    prompt_tokens=req.prompt_len,
                    e2e_ms=(now - req.arrival_t) * 1e3))
                self.waiting.remove(req)
                self.mgr.free(req.request_id)
                self._by_id.pop(req.request_id, None)
            non_emb = [r for r in finishing if not r.is_embedding]
            if non_emb:
                # mixed batch: recompute logits for the non-embedding rows
                rows = torch.tensor(
                    [seq_starts[i + 1] - 1 for i, (r, _) in enumerate(selected)
                     if r in non_emb], dtype=torch.int64, device=self.device)
                logits = hidden_or_logits[rows] @ self.model.lm_head.t()
                self._finish_prefills(non_emb, logits, outputs)
        else:
            if finishing:
                self._finish_prefills(finishing, hidden_or_logits, outputs)
        return outputs

    def _finish_prefills(self, finishing, logits, outputs) -> None:
        tokens = self._sample_host(logits, [r.temperature for r in finishing])
        now = time.time()
        for req, tok in zip(finishing, tokens):
            if not req.first_token_t:     # preempted reqs keep their TTFT
                req.first_token_t = now
                if self.ttft_slo_ms is not None and req.arrival_t and \
                        (now - req.arrival_t) * 1e3 > self.ttft_slo_ms:
                    req.slo_ok = False
            self.waiting.remove(req)
            if req.prefill_only:
                outputs.append(RequestOutput(
                    request_id=req.request_id, new_tokens=[int(tok)],
                    kind="prefill_done",
                    kv_blocks=list(self.mgr.tables[req.request_id]),
                    seq_len=self.mgr.seq_lens[req.request_id],
                    first_token=int(tok), prompt_tokens=req.prompt_len,
                    cached_tokens=req.cached_tokens,
                    ttft_ms=(now - req.arrival_t) * 1e3))
                # blocks stay allocated until the transfer engine releases
                self._by_id.pop(req.request_id, None)
            else:
                req.generated.append(int(tok))
                if req.slo_ok:
                    self.total_generated_slo += 1
                self.running.append(req)
                self.dstate.join(req, self.mgr.tables[req.request_id],
                                 self.mgr.seq_lens[req.request_id], int(tok))
                outputs.append(RequestOutput(
                    request_id=req.request_id, new_tokens=[int(tok)],
                    ttft_ms=(now - req.arrival_t) * 1e3,
                    cached_tokens=req.cached_tokens))
                self.total_generated += 1

    def release_prefilled(self, request_id: str) -> None:
        """Free a prefill-only sequence after its KV was shipped over xGMI."""
        self.mgr.free(request_id)

    # ---- decode ----
    def _decode_pass(self) -> List[RequestOutput]:
        """Assemble and launch step N, then collect step N-1's tokens.
        Membership and finishes are decided from host-side counts, so the
        launch never waits on token values."""
        st = self.dstate
        bs = self.pool.block_size
        active: List[EngineRequest] = []
        upd_rows: List[int] = []
        upd_idx: List[int] = []
        upd_val: List[int] = []
        done_now: List[EngineRequest] = []
        for req in self.running:
            if len(active) >= self.max_decode_batch:
                break
            if req.token_count >= req.max_tokens:
                if req.inflight == 0:
                    done_now.append(req)  # all tokens came from prefill
                continue          # else final token in flight; collect soon
            rid = req.request_id
            cur = self.mgr.seq_lens[rid]
            if cur % bs == 0 and len(self.mgr.tables[rid]) <= cur // bs:
                if not self.mgr.allocate(rid, cur + 1):
                    continue      # out of KV blocks: stall this sequence
                upd_rows.append(st.row_of[rid])
                upd_idx.append(cur // bs)
                upd_val.append(self.mgr.tables[rid][-1])
            self.mgr.set_seq_len(rid, cur + 1)
            req.inflight += 1
            active.append(req)

        finals = [self._finalize(r) for r in done_now]
        if not active:
            outs = self._collect_pending() + finals
            # KV exhaustion with every decodable sequence stalled: nothing
            # will ever free blocks — preempt the youngest running request
            # (vLLM-style recompute; the prefix cache usually resurrects
            # its just-freed blocks, so the recompute is mostly free)
            if self.running and self.mgr.free_blocks == 0:
                # victim: lowest InferenceObjective priority, youngest last
                victim = min(reversed(self.running),
                             key=lambda r: r.priority)
                po = self._preempt(victim)
                if po is not None:
                    outs.append(po)
            return outs

        if upd_rows:
            st.bt[torch.tensor(upd_rows, dtype=torch.int64, device=self.device),
                  torch.tensor(upd_idx, dtype=torch.int64, device=self.device)
                  ] = torch.tensor(upd_val, dtype=torch.int32,
                                   device=self.device)

        rows_t = st.rows_for([r.request_id for r in active])
        lens = st.seq_lens.index_select(0, rows_t)
        new_len = lens + 1
        pos = new_len - 1                           # int32 positions
        bt_rows = st.bt.index_select(0, rows_t)
        blk = bt_rows.gather(
            1, (pos // bs).to(torch.int64).unsqueeze(1)).squeeze(1)
        slot = blk.to(torch.int64) * bs + (pos % bs).to(torch.int64)
        st.seq_lens.index_copy_(0, rows_t, new_len)
        batch = ForwardBatch(
            input_ids=st.last_tok.index_select(0, rows_t),
            positions=pos, slot_mapping=slot, is_decode=True,
            block_tables=bt_rows, seq_lens=new_len,
            max_seq_len=max(self.mgr.seq_lens[r.request_id]
                            for r in active))
        logits = self.model.forward(batch, self.pool.tensor)
        any_temp = any(r.temperature > 0 for r in active)
        tok = self._sample_device(logits, rows_t, any_temp)
        st.last_tok.index_copy_(0, rows_t, tok)

        n = tok.shape[0]
        prev = self._collect_pending()     # read N-1 BEFORE reusing buffer
        if self._cuda:
            if self._pin is None or self._pin.shape[0] < st.capacity:
                self._pin = torch.empty(st.capacity, dtype=torch.int64,
                                        pin_memory=True)
            self._pin[:n].copy_(tok, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record()
            self._pending = (active, ev, n)
        else:
            self._pending = (active, None, tok)
        return prev + finals

    def _collect_pending(self) -> List[RequestOutput]:
        if self._pending is None:
            return []
        reqs, ev, payload = self._pending
        self._pending = None
        if ev is not None:
            ev.synchronize()
            vals = self._pin[:payload].tolist()
        else:
            vals = payload.tolist()
        outputs: List[RequestOutput] = []
        now = time.time()
        for req, tok in zip(reqs, vals):
            if self._by_id.get(req.request_id) is not req:
                continue                      # aborted while in flight
            req.inflight -= 1
            req.generated.append(int(tok))
            self.total_generated += 1
            if req.slo_ok:
                self.total_generated_slo += 1
            stopped = bool(req.stop_token_ids) and \
                int(tok) in req.stop_token_ids
            finished = stopped or len(req.generated) >= req.max_tokens
            out = RequestOutput(request_id=req.request_id,
                                new_tokens=[int(tok)], finished=finished,
                                finish_reason="stop" if stopped
                                else "length")
            if finished:
                n_gen = len(req.generated)
                out.all_tokens = list(req.generated)
                out.prompt_tokens = req.prompt_len
                out.completion_tokens = n_gen
                out.cached_tokens = req.cached_tokens
                out.ttft_ms = (req.first_token_t - req.arrival_t) * 1e3
                if n_gen > 1:
                    out.tpot_ms = (now - req.first_token_t) * 1e3 / (n_gen - 1)
                out.e2e_ms = (now - req.arrival_t) * 1e3
                self.running.remove(req)
                self.dstate.leave(req.request_id)
                self.mgr.free(req.request_id)
                self._by_id.pop(req.request_id, None)
            outputs.append(out)
        return outputs

    def _maybe_preempt_for(self, waiter: EngineRequest) -> bool:
        """Free KV for a strictly-higher-priority waiter by preempting the
        lowest-priority running request (youngest last) whose pending
        tokens are already collected. Returns True if a victim was
        preempted (its blocks free on the next step's allocation)."""
        # the pipelined decode keeps one token in flight on every running
        # request at prefill time: collect step N-1's tokens first so
        # victims become preemptible (their outputs surface via _carry)
        if any(r.inflight for r in self.running):
            self._carry.extend(self._collect_pending())
        candidates = [r for r in self.running
                      if r.inflight == 0 and r.priority < waiter.priority]
        if not candidates:
            return False
        victim = min(reversed(candidates), key=lambda r: r.priority)
        out = self._preempt(victim)
        if out is not None:
            self._rejects.append(out)   # surfaced next step()
        return True

    def _preempt(self, req: EngineRequest) -> Optional[RequestOutput]:
        """Evict a running request and requeue it for recompute: its
        generation so far folds into the prompt (prefill of prompt+generated
        produces the next token's logits, so accounting continues exactly).
        Pending tokens must be collected first (req.inflight == 0).

        If prompt+generated no longer fits the context window (can only
        happen for requests admitted before a config change — add_request
        clamps prompt+max_tokens to max_model_len), finish the request
        (finish_reason=length) instead of truncating: truncation would
        recompute KV that no longer corresponds to the retained generation
        and re-emit dropped tokens through all_tokens."""
        assert req.inflight == 0, "collect pending before preempting"
        merged = req.prompt_tokens + req.generated
        if len(merged) > self.max_model_len - 1:
            log.info("preempt would exceed context window; finishing",
                     req=req.request_id, generated=len(req.generated))
            return self._finalize(req)
        log.info("preempting for KV space", req=req.request_id,
                 generated=len(req.generated))
        self.running.remove(req)
        self.dstate.leave(req.request_id)
        self.mgr.free(req.request_id)
        req.prompt_tokens = merged
        req.computed = 0
        req.block_hashes = None
        req.registered_blocks = 0
        self.waiting.insert(0, req)
        return None

    def _finalize(self, req: EngineRequest) -> RequestOutput:
        """Emit the finished output for a request with no in-flight token
        (its full generation came from prefill, e.g. max_tokens=1)."""
        now = time.time()
        n_gen = len(req.generated)
        out = RequestOutput(
            request_id=req.request_id, new_tokens=[], finished=True,
            all_tokens=list(req.generated), prompt_tokens=req.prompt_len,
            completion_tokens=n_gen, cached_tokens=req.cached_tokens,
            ttft_ms=(req.first_token_t - req.arrival_t) * 1e3,
            e2e_ms=(now - req.arrival_t) * 1e3)
        if n_gen > 1:
            out.tpot_ms = (now - req.first_token_t) * 1e3 / (n_gen - 1)
        self.running.remove(req)
        self.dstate.leave(req.request_id)
        self.mgr.free(req.request_id)
        self._by_id.pop(req.request_id, None)
        return out

    # ---- sampling ----
    def _sample_device(self, logits: torch.Tensor, rows_t: torch.Tensor,
                       any_temp: bool) -> torch.Tensor:
        """Greedy argmax, or gumbel-max for temperature>0 rows — entirely on
        device (no host sync)."""
        if not any_temp:
            return logits.argmax(dim=-1)
        temps = self.dstate.temps.index_select(0, rows_t)
        hot = temps > 0
        inv = torch.where(hot, 1.0 / temps.clamp(min=1e-6),
                          torch.ones_like(temps))
        y = logits.float() * inv.unsqueeze(1)
        u = torch.rand_like(y).clamp_min(1e-20)
        gumbel = -torch.log(-torch.log(u).clamp_min(1e-20))
        y = y + gumbel * hot.unsqueeze(1)
        return y.argmax(dim=-1)

    def _sample_host(self, logits: torch.Tensor,
                     temps: List[float]) -> List[int]:
        """Prefill first-token sampling (host-visible; prefill already
        syncs for TTFT bookkeeping)."""
        if logits.numel() == 0:
            return []
        greedy = logits.argmax(dim=-1)
        if all(t <= 0 for t in temps):
            return greedy.tolist()
        out = []
        for i, t in enumerate(temps):
            if t <= 0:
                out.append(int(greedy[i]))
            else:
                p = torch.softmax(logits[i].float() / t, dim=-1)
                out.append(int(torch.multinomial(p, 1)))
        return out
