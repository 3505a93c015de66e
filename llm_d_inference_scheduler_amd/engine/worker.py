"""Per-GPU worker engine: continuous batching over the paged KV pool.

This is the MI355X replacement for the reference's external vLLM model-server
pods (the router only steers them). One EngineWorker owns
one GPU: chunked prefill + batched decode (gfx950 paged-attention kernel),
role-aware behavior (decode | prefill | encode | combinations), and the
vLLM-compatible metrics snapshot the datalayer collectors scrape.
"""
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

from ..datalayer.endpoint import Metrics, Role
from ..models.configs import ModelConfig
from ..models.llama import ForwardBatch, LlamaRunner
from ..utils.logging import get_logger
from .kvcache import BLOCK_SIZE, BlockManager, KVPool

log = get_logger("engine.worker")


@dataclass
class EngineRequest:
    request_id: str
    prompt_tokens: List[int]
    max_tokens: int = 16
    temperature: float = 0.0
    is_embedding: bool = False
    prefill_only: bool = False      # disagg: stop after prefill + 1st token
    cached_tokens: int = 0          # router-estimated prefix reuse (metrics)
    # multimodal: embeddings for the first len(prefix_embeds) prompt rows
    # (prompt_tokens must carry placeholder ids for those positions)
    prefix_embeds: "Optional[torch.Tensor]" = None
    # runtime state
    computed: int = 0               # prompt tokens already prefilled
    generated: List[int] = field(default_factory=list)
    slo_ok: bool = True             # TTFT met the SLO (goodput accounting)
    arrival_t: float = 0.0
    first_token_t: float = 0.0

    def __post_init__(self):
        if not self.arrival_t:
            self.arrival_t = time.time()

    @property
    def prompt_len(self) -> int:
        return len(self.prompt_tokens)


@dataclass
class RequestOutput:
    request_id: str
    new_tokens: List[int]
    finished: bool = False
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
                 seed: int = 0):
        self.cfg = config
        self.device = torch.device(device)
        self.role = role
        self.dtype = dtype
        self.prefill_chunk_tokens = prefill_chunk_tokens
        self.max_decode_batch = max_decode_batch
        self.max_model_len = max_model_len
        self.ttft_slo_ms = ttft_slo_ms
        if kv_blocks is None:
            kv_blocks = KVPool.blocks_for_budget(config, kv_budget_bytes,
                                                 dtype_bytes=2)
        self.pool = KVPool(config, kv_blocks, self.device, dtype)
        self.mgr = BlockManager(kv_blocks, self.pool.block_size)
        self.model = LlamaRunner(config, self.device, dtype, seed=seed)
        self.waiting: List[EngineRequest] = []
        self.running: List[EngineRequest] = []
        self._by_id: Dict[str, EngineRequest] = {}
        self.steps = 0
        self.total_generated = 0
        self.total_generated_slo = 0
        self.total_prefilled = 0

    # ------------------------------------------------------------------
    def add_request(self, req: EngineRequest) -> None:
        if len(req.prompt_tokens) >= self.max_model_len:
            req.prompt_tokens = req.prompt_tokens[:self.max_model_len - 1]
        if any(t >= self.cfg.vocab_size or t < 0
               for t in req.prompt_tokens):
            # defensive clamp: a mismatched tokenizer must not crash the
            # engine loop (ids fold into the vocab deterministically)
            req.prompt_tokens = [t % self.cfg.vocab_size
                                 for t in req.prompt_tokens]
        self.waiting.append(req)
        self._by_id[req.request_id] = req

    def admit_transferred(self, req: EngineRequest, local_blocks: List[int],
                          seq_len: int, first_token: int) -> None:
        """Adopt a prefilled sequence whose KV arrived over xGMI."""
        self.mgr.adopt(req.request_id, local_blocks, seq_len)
        req.computed = req.prompt_len
        req.generated = [first_token]
        if not req.first_token_t:
            req.first_token_t = time.time()
        if self.ttft_slo_ms is not None and req.arrival_t and \
                (req.first_token_t - req.arrival_t) * 1e3 > self.ttft_slo_ms:
            req.slo_ok = False
        self._by_id[req.request_id] = req
        self.running.append(req)

    def abort(self, request_id: str) -> None:
        req = self._by_id.pop(request_id, None)
        if req is None:
            return
        if req in self.waiting:
            self.waiting.remove(req)
        if req in self.running:
            self.running.remove(req)
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
        return bool(self.waiting or self.running)

    # ------------------------------------------------------------------
    def step(self) -> List[RequestOutput]:
        """One engine iteration: a chunked-prefill pass (if any waiting) and
        one decode pass over the running batch."""
        outputs: List[RequestOutput] = []
        if self.waiting:
            outputs.extend(self._prefill_pass())
        if self.running:
            outputs.extend(self._decode_pass())
        self.steps += 1
        return outputs

    # ---- prefill ----
    def _prefill_pass(self) -> List[RequestOutput]:
        budget = self.prefill_chunk_tokens
        selected: List[tuple] = []
        for req in list(self.waiting):
            if budget <= 0:
                break
            remaining = req.prompt_len - req.computed
            if remaining <= 0:
                self.waiting.remove(req)
                continue
            chunk = min(remaining, budget)
            if not self.mgr.can_allocate(req.computed + chunk) and not selected:
                # cannot even fit one: stall this step (waiting queue grows)
                break
            if not self.mgr.allocate(req.request_id, req.computed + chunk):
                break
            selected.append((req, chunk))
            budget -= chunk
        if not selected:
            return []

        input_ids, positions, slots = [], [], []
        seq_starts, ctx_lens, tables, logit_rows = [0], [], [], []
        embed_rows, embed_vals = [], []
        finishing: List[EngineRequest] = []
        for req, chunk in selected:
            start, end = req.computed, req.computed + chunk
            if req.prefix_embeds is not None and start < req.prefix_embeds.shape[0]:
                e_end = min(end, req.prefix_embeds.shape[0])
                base_row = seq_starts[-1]
                embed_rows.extend(range(base_row, base_row + (e_end - start)))
                embed_vals.append(req.prefix_embeds[start:e_end])
            input_ids.extend(req.prompt_tokens[start:end])
            positions.extend(range(start, end))
            slots.extend(self.mgr.slots_for_range(req.request_id, start, end))
            self.mgr.set_seq_len(req.request_id, end)
            ctx_lens.append(end)
            tables.append(torch.tensor(self.mgr.tables[req.request_id],
                                       dtype=torch.int32, device=self.device))
            seq_starts.append(seq_starts[-1] + chunk)
            if end == req.prompt_len:
                logit_rows.append(seq_starts[-1] - 1)
                finishing.append(req)
            req.computed = end
            self.total_prefilled += chunk

        batch = ForwardBatch(
            input_ids=torch.tensor(input_ids, dtype=torch.int64,
                                   device=self.device),
            positions=torch.tensor(positions, dtype=torch.int32,
                                   device=self.device),
            slot_mapping=torch.tensor(slots, dtype=torch.int64,
                                      device=self.device),
            is_decode=False, seq_starts=seq_starts, ctx_lens=ctx_lens,
            prefill_block_tables=tables,
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
        tokens = self._sample(logits, [r.temperature for r in finishing])
        now = time.time()
        for req, tok in zip(finishing, tokens):
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
        batch_reqs = self.running[:self.max_decode_batch]
        input_ids, positions, slots, seq_lens, tables = [], [], [], [], []
        active: List[EngineRequest] = []
        for req in batch_reqs:
            slot = self.mgr.append_token_slot(req.request_id)
            if slot is None:
                continue  # out of KV blocks: stall this sequence
            active.append(req)
            input_ids.append(req.generated[-1])
            positions.append(self.mgr.seq_lens[req.request_id] - 1)
            slots.append(slot)
            seq_lens.append(self.mgr.seq_lens[req.request_id])
            tables.append(self.mgr.tables[req.request_id])
        if not active:
            return []
        max_blocks = max(len(t) for t in tables)
        bt = torch.zeros((len(active), max_blocks), dtype=torch.int32)
        for i, t in enumerate(tables):
            bt[i, :len(t)] = torch.tensor(t, dtype=torch.int32)
        batch = ForwardBatch(
            input_ids=torch.tensor(input_ids, dtype=torch.int64,
                                   device=self.device),
            positions=torch.tensor(positions, dtype=torch.int32,
                                   device=self.device),
            slot_mapping=torch.tensor(slots, dtype=torch.int64,
                                      device=self.device),
            is_decode=True,
            block_tables=bt.to(self.device),
            seq_lens=torch.tensor(seq_lens, dtype=torch.int32,
                                  device=self.device))
        logits = self.model.forward(batch, self.pool.tensor)
        tokens = self._sample(logits, [r.temperature for r in active])
        outputs: List[RequestOutput] = []
        now = time.time()
        for req, tok in zip(active, tokens):
            req.generated.append(int(tok))
            self.total_generated += 1
            if req.slo_ok:
                self.total_generated_slo += 1
            finished = len(req.generated) >= req.max_tokens
            out = RequestOutput(request_id=req.request_id,
                                new_tokens=[int(tok)], finished=finished)
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
                self.mgr.free(req.request_id)
                self._by_id.pop(req.request_id, None)
            outputs.append(out)
        return outputs

    # ---- sampling ----
    def _sample(self, logits: torch.Tensor, temps: List[float]) -> List[int]:
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
