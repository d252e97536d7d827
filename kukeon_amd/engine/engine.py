"""The MI355X serving engine: continuous batching over a paged KV cache.

Design (MI355X-first, not a port — the reference eminwux/kukeon has no
inference plane at all):

* one engine per GPU process; sessions own persistent KV (multi-turn agent
  context stays resident in the 288 GB HBM across turns),
* prefill runs as a packed varlen batch (GEMM-bound at 32k tokens),
* decode runs the whole running set each step; the full forward + logits +
  top-k/top-p sampling is captured in a hipGraph per batch-size bucket so a
  ~2.5 ms bf16 decode step is not launch-bound (~300 kernel launches/step),
* requests hold *sticky rows* in the static decode buffers, so the per-step
  host work is O(active rows) scalar updates + one slab H2D copy — no
  per-step tensor rebuilds,
* split-KV decode attention (matrix-core kernel, 4 autonomous wave-slots
  per workgroup) is split to ~4096 wave-units so the chip stays full even
  at small batch; idle sessions' KV evicts under pressure (history
  recompute) and running rows preempt-by-recompute on exhaustion.
"""
from __future__ import annotations

import logging
import weakref
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch

from kukeon_amd import ops
from kukeon_amd.engine.config import EngineConfig, ModelConfig, SamplingParams
from kukeon_amd.engine.kv_cache import PagedKVCache, SequenceKV
from kukeon_amd.models.llama import AttnMeta

log = logging.getLogger("kukeon.engine")


@dataclass
class Request:
    req_id: int
    kv: SequenceKV                      # persistent, owned by the session
    prompt_tokens: List[int]            # new tokens to prefill this turn
    sampling: SamplingParams
    ctx_start: int = 0                  # kv.num_tokens at submission
    output_tokens: List[int] = field(default_factory=list)
    finished: bool = False
    row: int = -1                       # sticky decode row
    bt_written: int = 0                 # block-table entries already staged
    prefill_pos: int = 0                # prompt tokens already prefilled
    preempted: int = 0


@dataclass
class StepOutput:
    req_id: int
    new_tokens: List[int]
    finished: bool


class LLMEngine:
    def __init__(self, model, cfg: ModelConfig, ecfg: EngineConfig,
                 device="cuda"):
        self.model = model
        self.cfg = cfg
        self.ecfg = ecfg
        self.device = torch.device(device)
        self.is_cuda = self.device.type == "cuda"
        import kukeon_amd.parallel as parallel
        self.hk = cfg.num_kv_heads // max(1, parallel.tp_size())
        self.hq = cfg.num_q_heads // max(1, parallel.tp_size())

        kv_torch_dtype = (torch.uint8 if ecfg.kv_dtype == "fp8"
                          else torch.bfloat16)
        kv_bytes = 1 if ecfg.kv_dtype == "fp8" else 2
        nblocks = ecfg.num_kv_blocks
        if nblocks <= 0:
            if self.is_cuda:
                free, _total = torch.cuda.mem_get_info(self.device)
                # oversubscribed rehearsals (more local ranks than GPUs,
                # e.g. a world-2 torchrun on a 1-GPU box) share the HBM:
                # split the KV budget so the second rank isn't starved
                import os as _os
                lw = int(_os.environ.get("LOCAL_WORLD_SIZE", "1") or 1)
                ndev = max(1, torch.cuda.device_count())
                share = max(1, -(-lw // ndev))
                budget = int(free * ecfg.kv_mem_fraction) // share - (2 << 30)
                nblocks = PagedKVCache.blocks_from_bytes(
                    max(budget, 1 << 28), cfg.num_layers, self.hk,
                    ecfg.block_size, cfg.head_dim, dtype_bytes=kv_bytes)
                nblocks = min(nblocks, 4_000_000)
            else:
                nblocks = 512
        self.kv = PagedKVCache(cfg.num_layers, nblocks, self.hk,
                               ecfg.block_size, cfg.head_dim, self.device,
                               dtype=kv_torch_dtype)
        log.info("KV cache: %d blocks (%.1f GiB for K+V)", nblocks,
                 self.kv.k.nbytes * 2 / (1 << 30))

        self.waiting: List[Request] = []
        self._next_id = 0
        # every SequenceKV ever submitted (weak): idle-session KV is
        # evictable when admission starves on blocks (history recompute)
        self._known_kvs: "weakref.WeakValueDictionary[int, SequenceKV]" = \
            weakref.WeakValueDictionary()

        # ---- sticky-row decode state ----
        self.Bmax = ecfg.max_sessions
        self._rows: List[Optional[Request]] = [None] * self.Bmax
        self._free_rows: List[int] = list(range(self.Bmax - 1, -1, -1))
        self.num_running = 0

        mb = ecfg.max_blocks_per_seq
        d = self.device
        self.d_ids = torch.zeros(self.Bmax, dtype=torch.int32, device=d)
        self.d_pos = torch.zeros(self.Bmax, dtype=torch.int32, device=d)
        self.d_slots = torch.full((self.Bmax,), -1, dtype=torch.int32, device=d)
        self.d_bt = torch.zeros(self.Bmax, mb, dtype=torch.int32, device=d)
        self.d_seq_lens = torch.zeros(self.Bmax, dtype=torch.int32, device=d)
        self.d_temps = torch.zeros(self.Bmax, dtype=torch.float32, device=d)
        self.d_topk = torch.zeros(self.Bmax, dtype=torch.int32, device=d)
        self.d_topp = torch.ones(self.Bmax, dtype=torch.float32, device=d)
        self.d_tokens = torch.zeros(self.Bmax, dtype=torch.int32, device=d)
        self.d_seed = torch.full((1,), ecfg.seed, dtype=torch.int64, device=d)
        self.d_ws = torch.zeros(self.Bmax, 528, dtype=torch.float32,
                                device=d)  # sampling workspace rows
        # 32 split slots: small decode batches (16-session / 70B configs)
        # need B*Hk*S ~ 4096 wave-units to fill the MFMA attention kernel
        # at its 4-waves/SIMD occupancy; 16 capped them at half the chip
        self.max_splits = 32
        self.d_tmp_out = {}
        self.d_tmp_ml = {}
        # self-advancing decode: token ring + device step counter
        self.max_microbatch = 256
        self.d_ring = torch.zeros(self.max_microbatch * self.Bmax,
                                  dtype=torch.int32, device=d)
        self.d_step = torch.zeros(1, dtype=torch.int32, device=d)
        pin = self.is_cuda
        self.h_ids = torch.zeros(self.Bmax, dtype=torch.int32, pin_memory=pin)
        self.h_pos = torch.zeros(self.Bmax, dtype=torch.int32, pin_memory=pin)
        self.h_slots = torch.full((self.Bmax,), -1, dtype=torch.int32,
                                  pin_memory=pin)
        self.h_seq_lens = torch.zeros(self.Bmax, dtype=torch.int32,
                                      pin_memory=pin)
        self.h_bt = torch.zeros(self.Bmax, mb, dtype=torch.int32,
                                pin_memory=pin)
        self.h_params = torch.zeros(self.Bmax, 3, dtype=torch.float32,
                                    pin_memory=pin)
        self._params_dirty = False
        # numpy views of the pinned staging buffers: per-row scalar writes
        # through torch cost ~5-10us each; through numpy they are ~50ns,
        # and the views share the pinned memory so the H2D copies see them
        self.n_ids = self.h_ids.numpy()
        self.n_pos = self.h_pos.numpy()
        self.n_slots = self.h_slots.numpy()
        self.n_seq_lens = self.h_seq_lens.numpy()
        self.n_bt = self.h_bt.numpy()

        self.graphs: Dict[int, torch.cuda.CUDAGraph] = {}
        self._graph_pool = None

    # ------------------------------------------------------------------
    def add_request(self, kv: SequenceKV, prompt_tokens: List[int],
                    sampling: SamplingParams) -> int:
        rid = self._next_id
        self._next_id += 1
        self._known_kvs[id(kv)] = kv
        prompt = list(prompt_tokens)
        if kv.pending_token is not None:
            prompt.insert(0, kv.pending_token)
            kv.pending_token = None
        if kv.num_tokens == 0 and kv.history:
            # session was evicted while idle: rebuild the whole context
            # by prefilling its history ahead of the new tokens
            prompt = list(kv.history) + prompt
            kv.history = []
        if kv.num_tokens + len(prompt) + sampling.max_new_tokens > \
                self.ecfg.max_model_len:
            raise ValueError("request exceeds max_model_len")
        self.waiting.append(Request(rid, kv, prompt, sampling,
                                    ctx_start=kv.num_tokens))
        return rid

    def free_sequence(self, kv: SequenceKV) -> None:
        self.kv.allocator.free(kv.blocks)
        kv.blocks = []
        kv.num_tokens = 0
        kv.pending_token = None
        kv.history = []

    def has_work(self) -> bool:
        return bool(self.waiting) or self.num_running > 0

    # ------------------------------------------------------------------
    def step(self) -> List[StepOutput]:
        import kukeon_amd.parallel as parallel
        if parallel.ep_size() > 1:
            return self._step_ep_lockstep()
        return self._step_local()

    def _step_ep_lockstep(self) -> List[StepOutput]:
        """EP serving: every MoE forward is a collective, so all EP
        ranks must make the SAME number of model passes. Per lockstep
        round each rank contributes exactly ONE pass — a real step
        (decode micro-batch forced to 1) or a participation pass that
        only serves this rank's experts. The caller (modelhub engine
        loop) exchanges the has-work flag so idle ranks keep
        participating while any rank still serves."""
        from kukeon_amd.models.mixtral import MixtralModel
        self.ecfg.decode_microbatch = 1  # one pass per step, always
        before = MixtralModel.pass_count
        outs = self._step_local() if self.has_work() else []
        made = MixtralModel.pass_count - before
        assert made <= 1, f"EP lockstep violated: {made} passes in a step"
        if made == 0:
            self.model.participate()
        return outs

    def _step_local(self) -> List[StepOutput]:
        if self.waiting:
            batch, blocked = self._admit_prefill()
            # several small idle sessions may need to go before the head
            # request fits: keep evicting until admission unblocks or no
            # victims remain (single-evict crashed on spread-out
            # idle context even though capacity existed)
            while not batch and blocked and self._evict_idle_kv():
                batch, blocked = self._admit_prefill()
            if batch:
                return self._run_prefill(batch)
            if blocked and self.num_running == 0:
                # nothing running to make progress, nothing left to evict:
                # the head request alone cannot ever fit
                req = self.waiting[0]
                raise MemoryError(
                    f"request {req.req_id} cannot be admitted: needs more "
                    "KV blocks than the pool holds even with every idle "
                    "session evicted")
        if self.num_running > 0:
            return self._run_decode()
        return []

    def _evict_idle_kv(self) -> bool:
        """Evict the KV of an idle session (resident context, no active
        request) to unblock admission; its history stays so the next
        request on that session transparently re-prefills the context.
        Sessions idle in HBM are the norm for agent serving — without
        this, resident-but-idle context starves new admissions forever."""
        active = {id(r.kv) for r in self.waiting}
        active.update(id(r.kv) for r in self._rows if r is not None)
        victim = None
        for kv in self._known_kvs.values():
            if id(kv) in active or not kv.blocks:
                continue
            if victim is None or len(kv.blocks) > len(victim.blocks):
                victim = kv
        if victim is None:
            return False
        log.info("evicting idle session KV (%d blocks) to unblock admission",
                 len(victim.blocks))
        self.kv.allocator.free(victim.blocks)
        victim.blocks = []
        victim.num_tokens = 0
        return True

    # ------------------------------------------------------------------
    def _admit_prefill(self):
        """-> ([(request, chunk_len)], blocked_on_blocks); a chunk smaller
        than the remaining prompt keeps the request at the front of the
        queue (chunked prefill for prompts beyond the per-step token
        budget)."""
        batch: List[Tuple[Request, int]] = []
        blocked = False
        tokens = 0
        free = self.kv.allocator.num_free
        avail_rows = len(self._free_rows)
        budget = self.ecfg.max_prefill_tokens
        # soft tail: a request whose remainder only slightly overflows the
        # budget is admitted whole rather than leaving a tiny (<1/16
        # budget) chunk for a full extra model pass next step — at 64
        # sessions x 257 tokens vs a 16384 budget the hard cutoff cost an
        # entire 64-token full-layer sweep (~a decode step) per turn.
        slack = budget // 16
        while self.waiting and avail_rows > 0 and tokens < budget:
            req = self.waiting[0]
            remaining = len(req.prompt_tokens) - req.prefill_pos
            chunk = min(remaining, budget - tokens)
            if 0 < remaining - chunk <= slack and tokens + remaining <= \
                    budget + slack:
                chunk = remaining
            if chunk <= 0:
                break
            finishing = (req.prefill_pos + chunk) == len(req.prompt_tokens)
            reserve = chunk + (req.sampling.max_new_tokens if finishing
                               else 0)
            need = req.kv.blocks_needed(reserve)
            if need > free:
                blocked = True
                break
            free -= need
            tokens += chunk
            if finishing:
                avail_rows -= 1
                batch.append((self.waiting.pop(0), chunk))
            else:
                batch.append((req, chunk))
                break  # a partial chunk consumes the whole budget
        return batch, blocked

    def _alloc_for(self, req: Request, new_tokens: int) -> List[int]:
        need = req.kv.blocks_needed(new_tokens)
        if need:
            req.kv.blocks.extend(self.kv.allocator.alloc(need))
        slots = req.kv.slots_for(new_tokens)
        req.kv.num_tokens += new_tokens
        return slots

    # ------------------------------------------------------------------
    def _run_prefill(self, batch: List[Tuple[Request, int]]) -> List[StepOutput]:
        dev = self.device
        ids, pos, slots = [], [], []
        qs_pairs, qb_seq, qb_start = [], [], []
        row = 0
        for s, (req, chunk) in enumerate(batch):
            toks = req.prompt_tokens[req.prefill_pos:
                                     req.prefill_pos + chunk]
            start = req.kv.num_tokens
            sl = self._alloc_for(req, chunk)
            req.kv.history.extend(toks)
            ids.extend(toks)
            pos.extend(range(start, start + chunk))
            slots.extend(sl)
            qs_pairs.append((start, row))
            for qb in range(0, chunk, 32):
                qb_seq.append(s)
                qb_start.append(qb)
            row += chunk
            req.prefill_pos += chunk
        t_ids = torch.tensor(ids, dtype=torch.int32, device=dev)
        t_pos = torch.tensor(pos, dtype=torch.int32, device=dev)
        t_slots = torch.tensor(slots, dtype=torch.int32, device=dev)
        t_seq_lens = torch.tensor([r.kv.num_tokens for r, _ in batch],
                                  dtype=torch.int32, device=dev)
        t_qs = torch.tensor(qs_pairs, dtype=torch.int32, device=dev)
        mbt = max(len(r.kv.blocks) for r, _ in batch)
        bt = torch.zeros(len(batch), mbt, dtype=torch.int32)
        for i, (r, _) in enumerate(batch):
            bt[i, : len(r.kv.blocks)] = torch.tensor(r.kv.blocks,
                                                     dtype=torch.int32)
        meta = AttnMeta(mode="prefill", positions=t_pos, slot_mapping=t_slots,
                        block_table=bt.to(dev), seq_lens=t_seq_lens,
                        q_starts=t_qs,
                        qb_seq=torch.tensor(qb_seq, dtype=torch.int32,
                                            device=dev),
                        qb_start=torch.tensor(qb_start, dtype=torch.int32,
                                              device=dev))
        hidden = self.model.forward(t_ids, self.kv.k, self.kv.v, meta)
        # sample only the requests whose prompt completed this pass
        finishing = [(i, r) for i, (r, _) in enumerate(batch)
                     if r.prefill_pos == len(r.prompt_tokens)]
        if not finishing:
            return []
        last_rows, acc = [], 0
        for i, (r, chunk) in enumerate(batch):
            acc += chunk
            if r.prefill_pos == len(r.prompt_tokens):
                last_rows.append(acc - 1)
        logits = self.model.compute_logits(
            hidden[torch.tensor(last_rows, dtype=torch.long, device=dev)])
        toks = self._sample_eager([r for _, r in finishing], logits)
        return [self._append_token(r, t)
                for (_, r), t in zip(finishing, toks)]

    def _sample_eager(self, reqs: List[Request], logits: torch.Tensor):
        B = len(reqs)
        dev = self.device
        temps = torch.tensor([r.sampling.temperature for r in reqs],
                             dtype=torch.float32, device=dev)
        tk = torch.tensor([r.sampling.top_k for r in reqs], dtype=torch.int32,
                          device=dev)
        tp = torch.tensor([r.sampling.top_p for r in reqs],
                          dtype=torch.float32, device=dev)
        tokens = torch.zeros(B, dtype=torch.int32, device=dev)
        ops.sample(tokens, logits.contiguous(), temps, tk, tp, self.d_seed,
                   self.d_ws[:B])
        return tokens.cpu().tolist()

    # ------------------------------------------------------------------
    def _assign_row(self, req: Request) -> None:
        row = self._free_rows.pop()
        req.row = row
        req.bt_written = 0
        self._rows[row] = req
        self.num_running += 1
        self.h_params[row, 0] = req.sampling.temperature
        self.h_params[row, 1] = float(req.sampling.top_k)
        self.h_params[row, 2] = req.sampling.top_p
        self._params_dirty = True

    def _release_row(self, req: Request) -> None:
        if req.row >= 0:
            self._rows[req.row] = None
            self._free_rows.append(req.row)
            self.n_slots[req.row] = -1
            self.n_seq_lens[req.row] = 0
            req.row = -1
            self.num_running -= 1

    def cancel(self, rid: int) -> bool:
        """Abort a request at the next step boundary (client went away /
        explicit cancel verb). Must run on the engine thread (the server
        routes it through the submit queue). The session's KV keeps the
        already-computed prefix — the turn just ends early, exactly like
        a stop token at the last emitted position."""
        for i, r in enumerate(self.waiting):
            if r.req_id == rid:
                self.waiting.pop(i)
                r.finished = True
                if r.output_tokens:
                    r.kv.pending_token = r.output_tokens[-1]
                return True
        for r in self._rows:
            if r is not None and r.req_id == rid:
                r.finished = True
                if r.output_tokens:
                    r.kv.pending_token = r.output_tokens[-1]
                self._release_row(r)
                return True
        return False

    def _append_token(self, req: Request, tok: int) -> StepOutput:
        req.output_tokens.append(tok)
        if (len(req.output_tokens) >= req.sampling.max_new_tokens
                or tok in (req.sampling.stop_token_ids or ())):
            req.finished = True
            req.kv.pending_token = tok
            self._release_row(req)
        elif req.row < 0:
            self._assign_row(req)
        return StepOutput(req.req_id, [tok], req.finished)

    # ------------------------------------------------------------------
    def _bucket(self, b: int) -> int:
        for s in self.ecfg.graph_buckets:
            if b <= s:
                return min(s, self.Bmax)
        return self.Bmax

    def _decode_splits(self, bucket: int) -> int:
        # target ~8x the 256 CUs in workgroups: the micro-bench
        # (scripts/bench_paged_attn.py) shows 2048-4096 WGs runs 5-9%
        # faster than the 512-WG minimum (imbalance absorption across
        # the 8 XCDs), flat beyond
        # target ~4096 split-units: the MFMA decode kernel runs B*Hk*S
        # autonomous waves (4 per workgroup), and 4096 exactly fills the
        # chip at its 4-waves/SIMD occupancy (multiple of 4 required —
        # each workgroup's waves own consecutive split slots)
        want = (16 * 256 + bucket * self.hk - 1) // (bucket * self.hk)
        return max(4, min(self.max_splits, ((want + 3) // 4) * 4))

    def _tmp_for(self, bucket: int, splits: int):
        key = (bucket, splits)
        if key not in self.d_tmp_out:
            self.d_tmp_out[key] = torch.zeros(
                bucket, self.hq, splits, self.cfg.head_dim,
                dtype=torch.float32, device=self.device)
            self.d_tmp_ml[key] = torch.zeros(bucket, self.hq, splits, 2,
                                             dtype=torch.float32,
                                             device=self.device)
        return self.d_tmp_out[key], self.d_tmp_ml[key]

    def _preempt(self, req: Request) -> None:
        """Evict a running request: free its KV and re-queue it for full
        recompute (history + its un-processed last token become the new
        prompt; already-emitted output_tokens are preserved)."""
        log.warning("preempting request %d (KV exhausted)", req.req_id)
        self._release_row(req)
        history = list(req.kv.history)
        last = req.output_tokens[-1] if req.output_tokens else None
        self.kv.allocator.free(req.kv.blocks)
        req.kv.blocks = []
        req.kv.num_tokens = 0
        req.kv.history = []
        req.kv.pending_token = None
        req.prompt_tokens = history + ([last] if last is not None else [])
        req.prefill_pos = 0
        req.preempted += 1
        self.waiting.insert(0, req)

    def _run_decode(self) -> List[StepOutput]:
        """A micro-batch of k decode steps with ONE host sync: the graph is
        self-advancing (sampled tokens feed back on device, cursors advance
        in-kernel, tokens land in a ring buffer)."""
        active0 = [r for r in self._rows if r is not None]
        k = min(min(r.sampling.max_new_tokens - len(r.output_tokens)
                    for r in active0),
                max(1, self.ecfg.decode_microbatch), self.max_microbatch)
        # ---- alloc k tokens of KV per request (preempt on exhaustion) ----
        row = 0
        while row < self.Bmax:
            r = self._rows[row]
            if r is None:
                row += 1
                continue
            need = r.kv.blocks_needed(k)
            if need:
                try:
                    r.kv.blocks.extend(self.kv.allocator.alloc(need))
                except MemoryError:
                    actives = [x for x in self._rows if x is not None]
                    victim = actives[-1] if actives[-1] is not r else r
                    self._preempt(victim)
                    if victim is r:
                        row += 1
                    continue
            row += 1
        if self.num_running == 0:
            return []
        maxrow = max(i for i, r in enumerate(self._rows) if r is not None)
        nrows = maxrow + 1
        active = []
        for row in range(nrows):
            r = self._rows[row]
            if r is None:
                self.n_seq_lens[row] = 0
                continue
            active.append((row, r))
            start = r.kv.num_tokens
            self.n_ids[row] = r.output_tokens[-1]
            self.n_pos[row] = start
            self.n_seq_lens[row] = start + 1
            n = len(r.kv.blocks)
            if n > r.bt_written:
                self.n_bt[row, r.bt_written: n] = r.kv.blocks[r.bt_written:]
                r.bt_written = n
        use_graph = self.is_cuda and self.ecfg.use_graphs
        bucket = self._bucket(nrows) if use_graph else nrows
        for row in range(nrows):
            if self._rows[row] is None:
                self.n_pos[row] = -1   # rope kernel skips the KV write
        for row in range(nrows, bucket):
            self.n_seq_lens[row] = 0
            self.n_pos[row] = -1
        nb = bucket
        self.d_ids[:nb].copy_(self.h_ids[:nb], non_blocking=True)
        self.d_pos[:nb].copy_(self.h_pos[:nb], non_blocking=True)
        self.d_seq_lens[:nb].copy_(self.h_seq_lens[:nb], non_blocking=True)
        self.d_bt[:nb].copy_(self.h_bt[:nb], non_blocking=True)
        if self._params_dirty:
            self.d_temps.copy_(self.h_params[:, 0], non_blocking=True)
            self.d_topk.copy_(self.h_params[:, 1].to(torch.int32),
                              non_blocking=True)
            self.d_topp.copy_(self.h_params[:, 2], non_blocking=True)
            self._params_dirty = False
        self.d_step.zero_()
        if use_graph and bucket not in self.graphs:
            # capture AFTER the staging copies: the warmup forward then runs
            # on the real current state (its KV writes land exactly where
            # the first replay re-writes the same values), never on stale
            # block tables pointing at freed blocks.
            self._capture_safely(bucket)
        if use_graph:
            g = self.graphs[bucket]
            for _ in range(k):
                g.replay()
        else:
            for _ in range(k):
                self._decode_forward(bucket)
        slab = self.d_ring[: k * bucket].view(k, bucket)[:, :nrows].cpu()
        outs = []
        for row, r in active:
            toks = [int(t) for t in slab[:, row]]
            prev = r.output_tokens[-1]
            r.kv.history.extend([prev] + toks[:-1])
            r.kv.num_tokens += k
            fin = False
            emitted = []
            for i, t in enumerate(toks):
                emitted.append(t)
                fin = self._append_token(r, t).finished
                if fin and i + 1 < k:
                    # stop token mid-microbatch: the device already
                    # advanced KV/history for the discarded tail — roll
                    # the host counters back so the next turn's context
                    # ends at the stop token (staging re-derives
                    # pos/seq_lens from kv.num_tokens)
                    extra = k - (i + 1)
                    r.kv.num_tokens -= extra
                    del r.kv.history[len(r.kv.history) - extra:]
                    break
                if fin:
                    break
            outs.append(StepOutput(r.req_id, emitted, fin))
        return outs

    def _decode_forward(self, B: int) -> None:
        splits = self._decode_splits(B)
        tmp_out, tmp_ml = self._tmp_for(B, splits)
        meta = AttnMeta(
            mode="decode", positions=self.d_pos[:B],
            slot_mapping=self.d_slots[:B], block_table=self.d_bt[:B],
            seq_lens=self.d_seq_lens[:B], num_splits=splits,
            tmp_out=tmp_out, tmp_ml=tmp_ml)
        hidden = self.model.forward(self.d_ids[:B], self.kv.k, self.kv.v, meta)
        logits = self.model.compute_logits(hidden)
        ops.sample(self.d_tokens[:B], logits, self.d_temps[:B],
                   self.d_topk[:B], self.d_topp[:B], self.d_seed,
                   self.d_ws[:B])
        ops.decode_advance(self.d_ids[:B], self.d_pos[:B],
                           self.d_seq_lens[:B], self.d_tokens[:B],
                           self.d_ring, self.d_step)

    def _capture_safely(self, bucket: int) -> None:
        """Capture a bucket graph mid-serving: the warmup forward mutates
        the self-advancing cursors, so snapshot and restore them."""
        saved = (self.d_ids.clone(), self.d_pos.clone(),
                 self.d_seq_lens.clone(), self.d_step.clone())
        self._capture(bucket)
        self.d_ids.copy_(saved[0])
        self.d_pos.copy_(saved[1])
        self.d_seq_lens.copy_(saved[2])
        self.d_step.copy_(saved[3])
        if self.is_cuda:
            torch.cuda.synchronize()

    def _capture(self, bucket: int) -> None:
        log.info("capturing decode graph for bucket %d", bucket)
        torch.cuda.synchronize()
        self._decode_forward(bucket)  # warm up allocator + BLAS heuristics
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g, pool=self._graph_pool):
            self._decode_forward(bucket)
        if self._graph_pool is None:
            self._graph_pool = g.pool()
        self.graphs[bucket] = g
        torch.cuda.synchronize()

    def capture_all(self) -> None:
        """Pre-capture decode buckets before serving (safe buffer state)."""
        if not (self.is_cuda and self.ecfg.use_graphs):
            return
        self.d_slots.fill_(-1)
        self.d_seq_lens.zero_()
        self.d_ids.zero_()
        self.d_pos.fill_(-1)   # rope kernel skips KV writes for pos < 0
        for b in sorted(self.ecfg.graph_buckets, reverse=True):
            b = min(b, self.Bmax)
            if b not in self.graphs:
                self._capture(b)
