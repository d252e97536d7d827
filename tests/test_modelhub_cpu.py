"""Modelhub server tests on CPU (tiny model): concurrent sessions over the
unix socket, persistent multi-turn KV, release, and the system-cell
provisioning path through the controller + process runtime."""
import threading
import time
import uuid

import pytest

from kukeon_amd.engine.config import EngineConfig, tiny_llama
from kukeon_amd.models.llama import LlamaModel
from kukeon_amd.serve.server import ModelhubClient, ModelhubServer


@pytest.fixture
def hub():
    sock = f"/tmp/mh-{uuid.uuid4().hex[:10]}.sock"
    cfg = tiny_llama()
    ecfg = EngineConfig(max_model_len=256, max_sessions=8, num_kv_blocks=256,
                        use_graphs=False)
    model = LlamaModel(cfg, device="cpu")
    h = ModelhubServer(model, cfg, ecfg, sock, device="cpu")
    h.start()
    yield h, sock
    h.stop()


def test_generate_multi_turn_and_sessions(hub):
    h, sock = hub
    c = ModelhubClient(sock, timeout=120)
    r1 = c.generate("alice", [1, 2, 3, 4], max_new_tokens=4, temperature=0.0)
    assert len(r1["tokens"]) == 4
    assert r1["context_len"] == 4 + 4 - 1  # final token pending
    r2 = c.generate("alice", [9, 9], max_new_tokens=3, temperature=0.0)
    assert r2["context_len"] == r1["context_len"] + 1 + 2 + 3 - 1
    rb = c.generate("bob", [5, 6, 7], max_new_tokens=2, temperature=0.0)
    assert len(rb["tokens"]) == 2
    st = c.call("stats")
    assert st["sessions"] == 2
    c.call("release", session="alice")
    assert c.call("stats")["sessions"] == 1
    c.close()


def test_concurrent_clients_batched(hub):
    h, sock = hub
    results = {}

    def worker(name):
        c = ModelhubClient(sock, timeout=120)
        results[name] = c.generate(name, [ord(x) % 512 for x in name] * 3,
                                   max_new_tokens=5, temperature=0.0)
        c.close()

    ts = [threading.Thread(target=worker, args=(f"s{i}",)) for i in range(4)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=120)
    assert len(results) == 4
    assert all(len(r["tokens"]) == 5 for r in results.values())


def test_modelhub_as_system_cell(tmp_path):
    from kukeon_amd.controller.core import Controller
    from kukeon_amd.serve.server import ModelhubClient
    ctl = Controller(str(tmp_path / "run"), gpu_devices=[])
    ctl.bootstrap()
    sock = f"/tmp/mhc-{uuid.uuid4().hex[:8]}.sock"
    doc = ctl.provision_modelhub_cell(
        model="tiny-llama", gpus=0, socket_path=sock,
        extra_args=["--max-model-len", "128", "--kv-blocks", "64"])
    assert doc.status.state == "Ready"
    deadline = time.monotonic() + 60
    client = None
    while time.monotonic() < deadline:
        try:
            client = ModelhubClient(sock, timeout=60)
            break
        except OSError:
            time.sleep(0.5)
    assert client is not None, "modelhub cell never served its socket"
    r = client.generate("agent-1", [1, 2, 3], max_new_tokens=3,
                        temperature=0.0)
    assert len(r["tokens"]) == 3
    client.close()
    ctl.kill_cell("kuke-system", "kukeon", "kukeon", "modelhub")


def test_streaming_generate(tmp_path):
    """stream=True delivers incremental token deltas ahead of the final
    result, and the concatenated deltas equal the final token list."""
    import torch
    from kukeon_amd.engine.config import EngineConfig, tiny_llama
    from kukeon_amd.models.llama import LlamaModel
    from kukeon_amd.serve.server import ModelhubClient, ModelhubServer

    torch.manual_seed(0)
    cfg = tiny_llama()
    ecfg = EngineConfig(max_model_len=256, max_sessions=4, num_kv_blocks=128,
                        use_graphs=False, decode_microbatch=2)
    sock = f"/tmp/mhs-{uuid.uuid4().hex[:8]}.sock"
    hub = ModelhubServer(LlamaModel(cfg, device="cpu"), cfg, ecfg, sock,
                         device="cpu")
    hub.start()
    try:
        c = ModelhubClient(sock, timeout=120)
        deltas = []
        final = None
        for frame in c.generate_stream("s1", [5, 6, 7], max_new_tokens=9,
                                       temperature=0.0):
            if "delta" in frame:
                deltas.extend(frame["delta"])
                assert final is None
            else:
                final = frame
        assert final is not None and len(final["tokens"]) == 9
        assert deltas == final["tokens"][: len(deltas)]
        assert len(deltas) >= 2  # micro-batch 2 => incremental frames
        # non-streaming call still works on the same connection
        r = c.generate("s1", [9], max_new_tokens=3, temperature=0.0)
        assert len(r["tokens"]) == 3
        c.close()
    finally:
        hub.stop()


def test_release_refuses_while_request_in_flight(tmp_path):
    """Freeing a session's KV under an in-flight request would hand its
    blocks to another sequence mid-decode; release must refuse."""
    import threading
    import time as _time

    import torch

    from kukeon_amd.engine.config import EngineConfig, tiny_llama
    from kukeon_amd.models.llama import LlamaModel
    from kukeon_amd.serve.server import ModelhubClient, ModelhubServer

    torch.manual_seed(0)
    cfg = tiny_llama()
    ecfg = EngineConfig(max_model_len=256, max_sessions=4, num_kv_blocks=128,
                        use_graphs=False, decode_microbatch=1)
    sock = f"/tmp/mhr-{uuid.uuid4().hex[:8]}.sock"
    hub = ModelhubServer(LlamaModel(cfg, device="cpu"), cfg, ecfg, sock,
                         device="cpu")
    hub.start()
    try:
        c1 = ModelhubClient(sock, timeout=120)
        c2 = ModelhubClient(sock, timeout=120)
        t = threading.Thread(
            target=lambda: c1.generate("busy", [1, 2, 3],
                                       max_new_tokens=40, temperature=0.0))
        t.start()
        # wait until the request is actually running on the engine
        deadline = _time.monotonic() + 30
        while _time.monotonic() < deadline:
            if hub.engine.num_running > 0 or hub.engine.waiting:
                break
            _time.sleep(0.005)
        refused = False
        try:
            c2.call("release", session="busy")
        except RuntimeError as e:
            refused = "active request" in str(e)
        t.join(timeout=60)
        # either we raced past completion, or it refused while in flight
        assert refused or hub.engine.num_running == 0
        # after completion release succeeds
        c2.call("release", session="busy")
        assert hub.engine.kv.allocator.num_free == ecfg.num_kv_blocks
        c1.close(); c2.close()
    finally:
        hub.stop()


def test_server_concurrent_storm(tmp_path):
    """Thread storm against the server: interleaved generate/release/
    stats from many clients on a tiny KV pool (forces eviction) must
    neither deadlock nor corrupt accounting."""
    import random
    import threading

    import torch

    from kukeon_amd.engine.config import EngineConfig, tiny_llama
    from kukeon_amd.models.llama import LlamaModel
    from kukeon_amd.serve.server import ModelhubClient, ModelhubServer

    torch.manual_seed(0)
    cfg = tiny_llama()
    ecfg = EngineConfig(max_model_len=128, max_sessions=4, num_kv_blocks=24,
                        use_graphs=False, decode_microbatch=2)
    sock = f"/tmp/mhst-{uuid.uuid4().hex[:8]}.sock"
    hub = ModelhubServer(LlamaModel(cfg, device="cpu"), cfg, ecfg, sock,
                         device="cpu")
    hub.start()
    errs = []

    def worker(wid):
        rng = random.Random(wid)
        try:
            c = ModelhubClient(sock, timeout=120)
            for i in range(6):
                op = rng.choice(["gen", "gen", "gen", "rel", "stats"])
                sess = f"s{rng.randint(0, 5)}"
                if op == "gen":
                    r = c.generate(sess,
                                   [rng.randrange(cfg.vocab_size)
                                    for _ in range(rng.randint(2, 8))],
                                   max_new_tokens=rng.randint(2, 6),
                                   temperature=0.0)
                    assert r["tokens"]
                elif op == "rel":
                    try:
                        c.call("release", session=sess)
                    except RuntimeError as e:
                        assert ("active request" in str(e) or
                                "exceeds" in str(e)), e
                else:
                    st = c.call("stats")
                    assert st["kv_blocks_total"] == 24
            c.close()
        except Exception as e:  # noqa: BLE001
            errs.append((wid, repr(e)))

    ts = [threading.Thread(target=worker, args=(i,)) for i in range(6)]
    for t in ts:
        t.start()
    for t in ts:
        t.join(timeout=180)
    assert not any(t.is_alive() for t in ts), "storm deadlocked"
    allowed = ("exceeds max_model_len", "cannot be admitted")
    real = [e for e in errs if not any(a in e[1] for a in allowed)]
    assert not real, real
    hub.stop()


def test_engine_step_error_fails_pending_and_keeps_serving(hub):
    """A raising engine.step must not silently wedge the daemon: pending
    requests get an error frame and the loop keeps serving afterwards
    (ADVICE r01). MemoryError (admission-level) -> continue."""
    h, sock = hub
    orig_step = h.engine.step
    fired = {"n": 0}

    def boom():
        if fired["n"] == 0:
            fired["n"] += 1
            raise MemoryError("synthetic: cannot admit")
        return orig_step()

    h.engine.step = boom
    c = ModelhubClient(sock, timeout=30)
    import pytest as _pytest
    with _pytest.raises(Exception) as ei:
        c.generate("doomed", [1, 2, 3], max_new_tokens=2, temperature=0.0)
    assert "engine step failed" in str(ei.value)
    # the loop survived: a fresh request completes normally
    r = c.generate("alive", [4, 5, 6], max_new_tokens=2, temperature=0.0)
    assert len(r["tokens"]) == 2
    c.close()


def test_stats_turn_latency_percentiles(hub):
    """The stats verb reports p50/p95/p99 turn latency over the last
    N completed turns (SURVEY aux: turn/latency metrics)."""
    h, sock = hub
    c = ModelhubClient(sock, timeout=120)
    c.generate("lat-s1", [5, 6, 7], max_new_tokens=4)
    c.generate("lat-s1", [8], max_new_tokens=4)
    lat = c.call("stats")["turn_latency_ms"]
    assert lat["n"] >= 2
    assert lat["p50"] > 0
    assert lat["p50"] <= lat["p95"] <= lat["p99"]
    c.close()


def test_cancel_aborts_inflight_request(hub):
    """`cancel` ends a session's in-flight generate at the next step
    boundary: the blocked generate returns what was produced so far with
    canceled=True, and the session stays usable."""
    import threading

    h, sock = hub
    c = ModelhubClient(sock, timeout=120)
    got = {}

    def long_gen():
        c2 = ModelhubClient(sock, timeout=120)
        got["r"] = c2.call("generate", session="c1",
                           tokens=[1, 2, 3], max_new_tokens=200,
                           temperature=0.0)
        c2.close()

    t = threading.Thread(target=long_gen)
    t.start()
    # wait until the request is actually running
    for _ in range(200):
        if c.call("stats")["running"] or c.call("stats")["waiting"]:
            break
        import time as _t
        _t.sleep(0.02)
    r = c.call("cancel", session="c1")
    assert r["canceled"] == 1
    t.join(30)
    assert not t.is_alive()
    assert got["r"].get("canceled") is True
    assert len(got["r"]["tokens"]) < 200
    # session still serves
    r2 = c.generate("c1", [9], max_new_tokens=2, temperature=0.0)
    assert len(r2["tokens"]) == 2
    c.close()
