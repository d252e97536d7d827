"""The modelhub server: the agent-backing inference service.

Runs as a system-realm cell (provisioned like kukeond itself — SURVEY.md
§2.9 integration contract): one process per GPU serving the engine over a
unix socket. Agent sessions hold named server-side KV contexts that stay
resident in HBM across turns.

Wire protocol (newline JSON, one connection per client, requests may
interleave across clients — the engine continuously batches them):
  {"id": N, "method": "generate", "params": {"session": "s1",
      "tokens": [...], "max_new_tokens": 128, "temperature": 0.7,
      "top_k": 50, "top_p": 0.9}}
  -> {"id": N, "result": {"tokens": [...], "context_len": M}}
  other methods: ping, stats (KV occupancy + turn-latency
  percentiles), cancel (abort a session's in-flight turn), release
  (free a session's KV). Passing
  "stream": true on generate delivers incremental
  {"id": N, "delta": [tok, ...]} frames (one per decode micro-batch)
  ahead of the final result frame — time-to-first-token for agents.
"""
from __future__ import annotations

import argparse
import collections
import json
import logging
import os
import queue
import socket
import socketserver
import threading
import time
from dataclasses import dataclass
from typing import Dict, List, Optional

from kukeon_amd.engine.config import (EngineConfig, MODEL_PRESETS,
                                      SamplingParams)
from kukeon_amd.engine.engine import LLMEngine
from kukeon_amd.engine.kv_cache import SequenceKV

log = logging.getLogger("kukeon.modelhub")


@dataclass
class _Pending:
    req_id: int
    reply: "queue.Queue"
    tokens: List[int]
    stream: bool = False
    t0: float = 0.0
    session: str = ""


class ModelhubServer:
    def __init__(self, model, cfg, ecfg: EngineConfig, socket_path: str,
                 device: str = "cuda:0"):
        self.engine = LLMEngine(model, cfg, ecfg, device=device)
        self.socket_path = socket_path
        self.sessions: Dict[str, SequenceKV] = {}
        self._submit: "queue.Queue" = queue.Queue()
        self._pending: Dict[int, _Pending] = {}
        self._lock = threading.Lock()
        self._stop = threading.Event()
        self._srv = None
        self._threads = []
        self.metrics = {"tokens_generated": 0, "turns_completed": 0,
                        "engine_steps": 0, "busy_seconds": 0.0}
        # last-N turn latencies (seconds) for the stats percentiles —
        # the SURVEY aux-subsystem requirement (turn/latency metrics)
        self._turn_latency = collections.deque(maxlen=512)

    # ---- engine loop --------------------------------------------------
    def _engine_loop(self):
        self.engine.capture_all()
        from kukeon_amd import parallel
        ep = parallel.ep_size() > 1
        stopping = False
        while True:
            if self._stop.is_set():
                stopping = True
            if stopping and not ep:
                return
            # drain submissions (EP never blocks: the lockstep flag
            # exchange below must run every iteration so an idle rank
            # keeps serving its experts for the others)
            try:
                while True:
                    fn = self._submit.get(
                        timeout=(0.02 if ep else
                                 None if not self.engine.has_work() and
                                 self._submit.empty() else 0.0))
                    if fn is None:
                        stopping = True
                        if not ep:
                            return
                        break
                    fn()
            except queue.Empty:
                pass
            if ep:
                # cross-rank lockstep admission + consensus shutdown:
                # step while ANY rank has work (idle and even STOPPING
                # ranks keep contributing participation passes so the
                # others' requests complete); exit only when every rank
                # voted stop — an EP modelhub drains global work first.
                import torch as _torch
                import torch.distributed as _dist
                has = 1 if (not stopping and self.engine.has_work()) else 0
                flag = _torch.tensor(
                    [has, 1 if stopping else 0],
                    device=self.engine.device if self.engine.is_cuda
                    else "cpu")
                _dist.all_reduce(flag, op=_dist.ReduceOp.SUM)
                if int(flag[1].item()) == _dist.get_world_size():
                    return
                if int(flag[0].item()) == 0:
                    continue
                if stopping or not self.engine.has_work():
                    self.engine.model.participate()
                    continue
            elif not self.engine.has_work():
                continue
            import time as _time
            t0 = _time.perf_counter()
            try:
                outs = self.engine.step()
            except Exception as exc:  # noqa: BLE001
                # A raising step must never silently kill this thread: the
                # process would stay alive with every pending (and future)
                # request hung to its reply timeout and RestartPolicy never
                # firing. Fail all pending replies, then either keep
                # serving (admission-level MemoryError leaves the engine
                # consistent) or exit so restart_policy revives the cell.
                log.exception("engine.step failed: %s", exc)
                with self._lock:
                    for rid, p in list(self._pending.items()):
                        p.reply.put({"error": f"engine step failed: {exc}"})
                        del self._pending[rid]
                if isinstance(exc, MemoryError):
                    # drop the unadmittable head request(s); engine state
                    # is consistent, keep serving the rest
                    self.engine.waiting.clear()
                    continue
                log.critical("engine thread fatal; exiting for restart "
                             "policy")
                os._exit(70)
            self.metrics["busy_seconds"] += _time.perf_counter() - t0
            self.metrics["engine_steps"] += 1
            with self._lock:
                for o in outs:
                    self.metrics["tokens_generated"] += len(o.new_tokens)
                    p = self._pending.get(o.req_id)
                    if p is None:
                        continue
                    p.tokens.extend(o.new_tokens)
                    if p.stream and o.new_tokens and not o.finished:
                        p.reply.put({"delta": list(o.new_tokens)})
                    if o.finished:
                        self.metrics["turns_completed"] += 1
                        if p.t0:
                            self._turn_latency.append(
                                _time.perf_counter() - p.t0)
                        del self._pending[o.req_id]
                        if p.stream:
                            p.reply.put({"delta": list(o.new_tokens),
                                         "_final_stub": True})
                        p.reply.put({"tokens": p.tokens})

    # ---- request handling --------------------------------------------
    def handle(self, method: str, params: dict, emit=None) -> dict:
        """Dispatch one request. `emit`, when provided, receives
        incremental frames for streaming methods (written by the
        connection handler as `{"id": N, "delta": [tokens...]}` lines
        ahead of the final result frame)."""
        if method == "ping":
            return {"ok": True, "pid": os.getpid()}
        if method == "stats":
            kv = self.engine.kv
            lat = sorted(self._turn_latency)

            def pct(q: float) -> float:
                if not lat:
                    return 0.0
                return round(lat[min(len(lat) - 1,
                                     int(q * len(lat)))] * 1000, 1)

            return {
                "sessions": len(self.sessions),
                "running": self.engine.num_running,
                "waiting": len(self.engine.waiting),
                "kv_blocks_total": kv.num_blocks,
                "kv_blocks_free": kv.allocator.num_free,
                "turn_latency_ms": {"p50": pct(0.50), "p95": pct(0.95),
                                    "p99": pct(0.99),
                                    "n": len(lat)},
                **self.metrics,
            }
        if method == "cancel":
            name = params["session"]
            done: "queue.Queue" = queue.Queue()

            def _do_cancel():
                with self._lock:
                    rids = [rid for rid, p in self._pending.items()
                            if p.session == name]
                n = 0
                for rid in rids:
                    if self.engine.cancel(rid):
                        n += 1
                    with self._lock:
                        p = self._pending.pop(rid, None)
                    if p is not None:
                        p.reply.put({"canceled": True,
                                     "tokens": p.tokens})
                done.put({"canceled": n})
            self._submit.put(_do_cancel)
            return done.get(timeout=60)
        if method == "release":
            name = params["session"]
            done: "queue.Queue" = queue.Queue()

            def _do_release():
                kv = self.sessions.get(name)
                if kv is None:
                    done.put({})
                    return
                busy = any(r is not None and r.kv is kv
                           for r in self.engine._rows) or \
                    any(r.kv is kv for r in self.engine.waiting)
                if busy:
                    # freeing blocks under an in-flight request would let
                    # the decode graph write into reallocated memory
                    done.put({"error": f"session {name} has an active "
                                       "request; release after it finishes"})
                    return
                self.sessions.pop(name, None)
                self.engine.free_sequence(kv)
                done.put({})
            self._submit.put(_do_release)
            out = done.get(timeout=60)
            if "error" in out:
                raise ValueError(out["error"])
            return out
        if method == "generate":
            return self._generate(params, emit)
        raise ValueError(f"unknown method {method}")

    def _generate(self, params: dict, emit=None) -> dict:
        name = params["session"]
        tokens = list(params["tokens"])
        stream = bool(params.get("stream", False)) and emit is not None
        sp = SamplingParams(
            temperature=float(params.get("temperature", 0.7)),
            top_k=int(params.get("top_k", 50)),
            top_p=float(params.get("top_p", 0.9)),
            max_new_tokens=int(params.get("max_new_tokens", 128)),
            stop_token_ids=tuple(params.get("stop_token_ids", ()) or ()))
        reply: "queue.Queue" = queue.Queue()

        def _do_submit():
            kv = self.sessions.get(name)
            if kv is None:
                kv = SequenceKV(self.engine.ecfg.block_size)
                self.sessions[name] = kv
            try:
                rid = self.engine.add_request(kv, tokens, sp)
            except Exception as e:  # overlong context etc.
                reply.put({"error": str(e)})
                return
            with self._lock:
                self._pending[rid] = _Pending(rid, reply, [], stream=stream,
                                              t0=time.perf_counter(),
                                              session=name)
        self._submit.put(_do_submit)
        while True:
            out = reply.get(timeout=600)
            if "error" in out:
                raise ValueError(out["error"])
            if "delta" in out:
                if not out.get("_final_stub"):
                    try:
                        emit({"delta": out["delta"]})
                    except OSError:
                        # streaming client went away: stop decoding for
                        # it at the next step boundary instead of
                        # burning GPU to max_new_tokens
                        self.handle("cancel", {"session": name})
                        raise
                continue
            break
        kv = self.sessions.get(name)
        out["context_len"] = kv.num_tokens if kv else 0
        return out

    # ---- server plumbing ----------------------------------------------
    def start(self):
        sp = self.socket_path
        if os.path.exists(sp):
            os.unlink(sp)
        os.makedirs(os.path.dirname(sp) or ".", exist_ok=True)
        hub = self

        class Handler(socketserver.StreamRequestHandler):
            def handle(self):
                for line in self.rfile:
                    line = line.strip()
                    if not line:
                        continue
                    try:
                        req = json.loads(line)
                        rid = req.get("id")

                        def emit(frame, _rid=rid):
                            self.wfile.write((json.dumps(
                                {"id": _rid, **frame}) + "\n").encode())
                            self.wfile.flush()
                        result = hub.handle(req.get("method", ""),
                                            req.get("params") or {},
                                            emit=emit)
                        resp = {"id": rid, "result": result}
                    except Exception as e:  # noqa: BLE001
                        resp = {"id": req.get("id") if isinstance(req, dict)
                                else None, "error": str(e)}
                    try:
                        self.wfile.write((json.dumps(resp) + "\n").encode())
                        self.wfile.flush()
                    except OSError:
                        return

        cwd = os.getcwd()
        bind_path = sp
        if len(sp) > 100:
            os.chdir(os.path.dirname(sp))
            bind_path = os.path.basename(sp)
        try:
            self._srv = socketserver.ThreadingUnixStreamServer(
                bind_path, Handler)
        finally:
            os.chdir(cwd)
        self._srv.daemon_threads = True
        # same ownership model as the kukeond socket: kukeon-group 0660
        # when the group exists, 0666 single-user fallback
        from kukeon_amd.runtime import sysuser
        gid = sysuser.lookup_group()
        if gid is not None:
            sysuser.apply_socket_group(sp, gid)
        else:
            os.chmod(sp, 0o666)
        t = threading.Thread(target=self._srv.serve_forever, daemon=True)
        t.start()
        self._threads.append(t)
        te = threading.Thread(target=self._engine_loop, daemon=True)
        te.start()
        self._threads.append(te)
        log.info("modelhub serving on %s", sp)

    def stop(self):
        self._stop.set()
        self._submit.put(None)
        if self._srv:
            self._srv.shutdown()
            self._srv.server_close()


class ModelhubClient:
    def __init__(self, socket_path: str, timeout: float = 600.0):
        self.sock = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        self.sock.settimeout(timeout)
        self.sock.connect(socket_path)
        self._rf = self.sock.makefile("rb")
        self._id = 0
        self._lock = threading.Lock()

    def call(self, method: str, **params):
        with self._lock:
            self._id += 1
            self.sock.sendall((json.dumps(
                {"id": self._id, "method": method, "params": params}) +
                "\n").encode())
            line = self._rf.readline()
        if not line:
            raise ConnectionError("modelhub closed the connection")
        resp = json.loads(line)
        if "error" in resp:
            raise RuntimeError(resp["error"])
        return resp["result"]

    def generate(self, session: str, tokens: List[int], **kw):
        return self.call("generate", session=session, tokens=tokens, **kw)

    def generate_stream(self, session: str, tokens: List[int], **kw):
        """Yield token deltas as the engine produces them; the final
        yield is the full result dict (with "tokens" and "context_len")."""
        with self._lock:
            self._id += 1
            self.sock.sendall((json.dumps(
                {"id": self._id, "method": "generate",
                 "params": {"session": session, "tokens": tokens,
                            "stream": True, **kw}}) + "\n").encode())
            while True:
                line = self._rf.readline()
                if not line:
                    raise ConnectionError("modelhub closed the connection")
                resp = json.loads(line)
                if "error" in resp:
                    raise RuntimeError(resp["error"])
                if "delta" in resp:
                    yield {"delta": resp["delta"]}
                    continue
                yield resp["result"]
                return

    def close(self):
        self.sock.close()


def main(argv: Optional[List[str]] = None) -> None:
    ap = argparse.ArgumentParser("kukeon-modelhub")
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--socket", default="/run/kukeon/modelhub.sock")
    ap.add_argument("--device", default=None)
    ap.add_argument("--max-model-len", type=int, default=4096)
    ap.add_argument("--max-sessions", type=int, default=256)
    ap.add_argument("--no-graphs", action="store_true")
    ap.add_argument("--kv-blocks", type=int, default=0)
    args = ap.parse_args(argv)
    from kukeon_amd.utils import logging as klog
    klog.setup(logging.INFO)
    import torch
    device = args.device or ("cuda:0" if torch.cuda.is_available() else "cpu")
    cfg = MODEL_PRESETS[args.model]()
    ecfg = EngineConfig(
        max_model_len=args.max_model_len, max_sessions=args.max_sessions,
        num_kv_blocks=args.kv_blocks or (512 if device == "cpu" else 0),
        use_graphs=not args.no_graphs and device != "cpu" and not cfg.is_moe)
    if cfg.is_moe:
        from kukeon_amd.models.mixtral import MixtralModel
        model = MixtralModel(cfg, device=device)
    else:
        from kukeon_amd.models.llama import LlamaModel
        model = LlamaModel(cfg, device=device)
    hub = ModelhubServer(model, cfg, ecfg, args.socket, device=device)
    hub.start()
    import signal
    stop = threading.Event()
    signal.signal(signal.SIGTERM, lambda *_: stop.set())
    signal.signal(signal.SIGINT, lambda *_: stop.set())
    while not stop.is_set():
        stop.wait(1.0)
    hub.stop()


if __name__ == "__main__":
    main()
