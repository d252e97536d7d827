"""Synthetic agent-session workload driver for the serving engine.

Models the BASELINE.json metric: N concurrent agent Sessions, each running
turns of (prompt tokens -> decode tokens) against a persistent KV context
that survives across turns (the paged cache keeps it resident in HBM);
when a session's context would exceed its cap the context is compacted
(freed + fresh system prompt), as agent runtimes do.
"""
from __future__ import annotations

import random
import time
from dataclasses import dataclass
from typing import Dict, List, Optional

from kukeon_amd.engine.config import SamplingParams
from kukeon_amd.engine.engine import LLMEngine
from kukeon_amd.engine.kv_cache import SequenceKV


@dataclass
class TurnStats:
    session_id: int
    submit_ts: float
    finish_ts: float = 0.0

    @property
    def latency(self) -> float:
        return self.finish_ts - self.submit_ts


class AgentSession:
    def __init__(self, sid: int, engine: LLMEngine, vocab: int,
                 first_prompt: int = 512, followup_prompt: int = 256,
                 decode_len: int = 128, ctx_cap: int = 3584,
                 sampling: Optional[SamplingParams] = None, seed: int = 0):
        self.sid = sid
        self.engine = engine
        self.vocab = vocab
        self.first_prompt = first_prompt
        self.followup_prompt = followup_prompt
        self.decode_len = decode_len
        self.ctx_cap = ctx_cap
        self.sampling = sampling or SamplingParams(max_new_tokens=decode_len)
        self.rng = random.Random(seed * 7919 + sid)
        self.kv = SequenceKV(engine.ecfg.block_size)
        self.turns_done = 0
        self.pending_rid: Optional[int] = None

    def _prompt(self, n: int) -> List[int]:
        return [self.rng.randrange(self.vocab) for _ in range(n)]

    def start_turn(self) -> int:
        """Submit one turn; returns the request id."""
        # effective context: cached tokens, or — if the engine evicted this
        # idle session's KV under pressure — the history add_request would
        # transparently re-prefill (plus the pending sampled token)
        ctx = self.kv.num_tokens or (
            len(self.kv.history) + (1 if self.kv.pending_token is not None
                                    else 0))
        n = self.first_prompt if ctx == 0 else self.followup_prompt
        if ctx + n + self.decode_len > self.ctx_cap:
            # context compaction: drop history, re-seed with a fresh prompt
            self.engine.free_sequence(self.kv)
            n = self.first_prompt
        rid = self.engine.add_request(self.kv, self._prompt(n), self.sampling)
        self.pending_rid = rid
        return rid


class TurnDriver:
    """Lockstep turn driver: every session runs exactly one turn per round."""

    def __init__(self, engine: LLMEngine, num_sessions: int, vocab: int,
                 **session_kw):
        self.engine = engine
        self.sessions = [AgentSession(i, engine, vocab, **session_kw)
                         for i in range(num_sessions)]
        self.turn_latencies: List[float] = []

    def run_round(self) -> int:
        """Run one round (each session completes one turn); returns #turns."""
        pending: Dict[int, TurnStats] = {}
        now = time.perf_counter()
        by_rid: Dict[int, AgentSession] = {}
        for s in self.sessions:
            rid = s.start_turn()
            pending[rid] = TurnStats(s.sid, now)
            by_rid[rid] = s
        done = 0
        while pending:
            outs = self.engine.step()
            t = time.perf_counter()
            for o in outs:
                if o.finished and o.req_id in pending:
                    st = pending.pop(o.req_id)
                    st.finish_ts = t
                    self.turn_latencies.append(st.latency)
                    sess = by_rid[o.req_id]
                    sess.turns_done += 1
                    done += 1
        return done
