"""Mailbox unit semantics (parallel/mailbox.py): pipelined one-step-deep
exchange, quiesce vs drain, sync fallback — with the collective stubbed
so ordering is observable without a process group."""
import threading
import time

from llm_d_inference_scheduler_amd.parallel.mailbox import Mailbox


def _stubbed(mb, log, delay=0.0):
    def fake_gather(outbox):
        if delay:
            time.sleep(delay)
        log.append(list(outbox))
        return [{"echo": m} for m in outbox]
    mb._gather = fake_gather
    return mb


class TestPipelinedMailbox:
    def test_returns_previous_exchange(self):
        log = []
        mb = _stubbed(Mailbox(object(), 0, 2, pipelined=True), log)
        assert mb.exchange([{"a": 1}]) == []            # nothing yet
        out = mb.exchange([{"b": 2}])
        assert out == [{"echo": {"a": 1}}]              # one step behind
        assert mb.drain() == [{"echo": {"b": 2}}]

    def test_one_collective_per_exchange_in_order(self):
        log = []
        mb = _stubbed(Mailbox(object(), 0, 2, pipelined=True), log)
        for i in range(5):
            mb.exchange([{"i": i}])
        mb.drain()
        assert log == [[{"i": i}] for i in range(5)]

    def test_quiesce_waits_without_consuming(self):
        log = []
        mb = _stubbed(Mailbox(object(), 0, 2, pipelined=True), log,
                      delay=0.05)
        mb.exchange([{"x": 1}])
        mb.quiesce()
        assert log == [[{"x": 1}]]                      # completed
        # the result is still delivered by the NEXT exchange
        assert mb.exchange([]) == [{"echo": {"x": 1}}]

    def test_blocked_time_accounting(self):
        log = []
        mb = _stubbed(Mailbox(object(), 0, 2, pipelined=True), log,
                      delay=0.03)
        mb.exchange([{"x": 1}])
        mb.exchange([{"y": 2}])     # must wait for the slow first gather
        assert mb.blocked_s > 0.0
        mb.drain()

    def test_world1_is_loopback(self):
        mb = Mailbox(None, 0, 1, pipelined=True)
        assert not mb.pipelined     # world 1 never pipelines
        assert mb.exchange([{"m": 1}]) == [{"m": 1}]

    def test_sync_mode_same_step(self):
        log = []
        mb = _stubbed(Mailbox(object(), 0, 2, pipelined=False), log)
        assert mb.exchange([{"a": 1}]) == [{"echo": {"a": 1}}]

    def test_for_me_filters_dst(self):
        mb = Mailbox(None, 1, 2)
        msgs = [{"dst": 0, "v": "x"}, {"dst": 1, "v": "y"}, {"v": "z"}]
        assert mb.for_me(msgs) == [{"dst": 1, "v": "y"}]
