"""Tensor-parallel correctness on CPU (gloo, world_size 2) — the multi-GPU
path is correct by construction and covered here without hardware; the
driver runs the real RCCL scaling bench at round end."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp
import torch.nn.functional as F


def _tp_linear_worker(rank, world, port, result_dir):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from kukeon_amd import parallel
    parallel._TP_GROUP = None
    parallel._TP_RANK = rank
    parallel._TP_SIZE = world

    torch.manual_seed(7)  # identical full weights on every rank
    T, H, I = 5, 64, 32
    x = torch.randn(T, H)
    Wg = torch.randn(I, H) * 0.1
    Wu = torch.randn(I, H) * 0.1
    Wd = torch.randn(H, I) * 0.1

    # reference (single-rank) computation
    ref = F.silu(x @ Wg.T) * (x @ Wu.T) @ Wd.T

    # TP: column-shard gate/up, row-shard down, one all-reduce
    il = I // world
    Wg_l, Wu_l = Wg[rank * il:(rank + 1) * il], Wu[rank * il:(rank + 1) * il]
    Wd_l = Wd[:, rank * il:(rank + 1) * il]
    act = F.silu(x @ Wg_l.T) * (x @ Wu_l.T)
    y = act @ Wd_l.T
    parallel.tp_all_reduce(y)
    torch.testing.assert_close(y, ref, rtol=1e-4, atol=1e-4)
    dist.destroy_process_group()


def _tp_engine_worker(rank, world, port, result_dir):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from kukeon_amd import parallel
    parallel.init_tensor_parallel(world)
    from kukeon_amd.engine.config import (EngineConfig, SamplingParams,
                                          tiny_llama)
    from kukeon_amd.engine.engine import LLMEngine
    from kukeon_amd.engine.kv_cache import SequenceKV
    from kukeon_amd.models.llama import LlamaModel

    cfg = tiny_llama()  # 2 q heads / 2 kv heads: shards 1+1 across tp=2
    ecfg = EngineConfig(max_model_len=128, max_sessions=2, num_kv_blocks=64,
                        use_graphs=False, tp_size=world)
    model = LlamaModel(cfg, device="cpu")
    engine = LLMEngine(model, cfg, ecfg, device="cpu")
    kv = SequenceKV(ecfg.block_size)
    engine.add_request(kv, [3, 1, 4, 1, 5, 9, 2, 6],
                       SamplingParams(temperature=0.0, max_new_tokens=5))
    toks = []
    while engine.has_work():
        for o in engine.step():
            toks.extend(o.new_tokens)
    assert len(toks) == 5
    # every TP rank must sample the identical token stream
    gathered = [None] * world
    dist.all_gather_object(gathered, toks)
    assert gathered[0] == gathered[1], gathered
    dist.destroy_process_group()


def _tp_overlap_worker(rank, world, port, result_dir):
    """Pipelined TP prefill (comm/GEMM overlap, 2 sequence groups) must
    produce exactly the tokens of the plain TP path."""
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from kukeon_amd import parallel
    parallel.init_tensor_parallel(world)
    from kukeon_amd.engine.config import (EngineConfig, SamplingParams,
                                          tiny_llama)
    from kukeon_amd.engine.engine import LLMEngine
    from kukeon_amd.engine.kv_cache import SequenceKV
    from kukeon_amd.models.llama import LlamaModel

    cfg = tiny_llama()
    prompts = [[3, 1, 4, 1, 5, 9, 2, 6], [2, 7, 1, 8, 2, 8],
               [1, 1, 2, 3, 5, 8, 13, 21, 34, 55]]
    runs = {}
    for mode in ("0", "1"):
        os.environ["KUKEON_TP_OVERLAP"] = mode
        ecfg = EngineConfig(max_model_len=128, max_sessions=4,
                            num_kv_blocks=96, use_graphs=False,
                            tp_size=world)
        model = LlamaModel(cfg, device="cpu")
        engine = LLMEngine(model, cfg, ecfg, device="cpu")
        for p in prompts:  # one multi-sequence prefill batch
            engine.add_request(SequenceKV(ecfg.block_size), list(p),
                               SamplingParams(temperature=0.0,
                                              max_new_tokens=4))
        toks = []
        while engine.has_work():
            for o in engine.step():
                toks.append((o.req_id, tuple(o.new_tokens)))
        runs[mode] = sorted(toks)
    assert runs["0"] == runs["1"], runs
    assert LlamaModel._overlap_runs > 0  # the pipelined path really ran
    del os.environ["KUKEON_TP_OVERLAP"]
    dist.destroy_process_group()


@pytest.mark.parametrize("worker", [_tp_linear_worker, _tp_engine_worker,
                                    _tp_overlap_worker])
def test_tp_world2(worker, tmp_path):
    port = 29600 + (os.getpid() + (0 if worker is _tp_linear_worker else 7)) % 500
    mp.spawn(worker, args=(2, port, str(tmp_path)), nprocs=2, join=True)


def _ep_moe_worker(rank, world, port, result_dir):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from kukeon_amd import parallel
    from kukeon_amd.engine.config import tiny_mixtral
    from kukeon_amd.models.mixtral import MixtralMoE

    cfg = tiny_mixtral()

    # single-rank reference module (seed S, all experts local)
    torch.manual_seed(31)
    ref_moe = MixtralMoE(cfg, "cpu")

    # expert-parallel module from the SAME rng stream -> same model
    parallel.init_expert_parallel(world)
    torch.manual_seed(31)
    ep_moe = MixtralMoE(cfg, "cpu")
    assert ep_moe.e_local == cfg.num_experts // world
    torch.testing.assert_close(
        ep_moe.gate_up_w,
        ref_moe.gate_up_w[rank * ep_moe.e_local:(rank + 1) * ep_moe.e_local])

    # each rank routes ITS OWN tokens; results must match the single-rank
    # module on those tokens exactly (same experts, same math)
    torch.manual_seed(100 + rank)
    x = torch.randn(7 + rank, cfg.hidden_size, dtype=torch.bfloat16)
    out_ep = ep_moe._forward_ep(x.clone())
    out_ref = ref_moe._forward_sparse(x.clone())
    torch.testing.assert_close(out_ep.float(), out_ref.float(), rtol=3e-2,
                               atol=3e-2)
    parallel.init_expert_parallel(1)
    dist.destroy_process_group()


def test_expert_parallel_moe_all_to_all(tmp_path):
    """EP MoE (BASELINE config 5 wording: expert all-to-all) matches the
    single-rank sparse path on every rank's tokens — gloo world 2."""
    port = 29600 + (os.getpid() + 13) % 500
    mp.spawn(_ep_moe_worker, args=(2, port, str(tmp_path)), nprocs=2,
             join=True)


def _ep_serving_worker(rank, world, port, result_dir):
    """EP SERVING: two modelhub servers, one per EP
    rank, under ASYMMETRIC load — rank 1 goes idle while rank 0 still
    serves; the lockstep scheduler keeps rank 1 participating (serving
    its experts) so rank 0's requests complete."""
    import uuid
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from kukeon_amd import parallel
    parallel.init_expert_parallel(world)
    from kukeon_amd.engine.config import EngineConfig, tiny_mixtral
    from kukeon_amd.models.mixtral import MixtralModel
    from kukeon_amd.serve.server import ModelhubClient, ModelhubServer

    cfg = tiny_mixtral()
    ecfg = EngineConfig(max_model_len=128, max_sessions=4,
                        num_kv_blocks=64, use_graphs=False)
    model = MixtralModel(cfg, device="cpu")
    sock = f"/tmp/mh-ep-{rank}-{uuid.uuid4().hex[:8]}.sock"
    hub = ModelhubServer(model, cfg, ecfg, sock, device="cpu")
    hub.start()
    try:
        c = ModelhubClient(sock, timeout=180)
        if rank == 1:
            # one short request, then idle while rank 0 keeps serving
            r = c.generate("s1", [5, 6, 7], max_new_tokens=2,
                           temperature=0.0)
            assert len(r["tokens"]) == 2
        else:
            for turn in range(3):
                r = c.generate("s0", [1 + turn, 2, 3], max_new_tokens=4,
                               temperature=0.0)
                assert len(r["tokens"]) == 4
        c.close()
    finally:
        # no main-thread barrier here: it would collide with the engine
        # thread's collectives. hub.stop() votes stop through the
        # lockstep flag; the engine loop exits only when EVERY rank has
        # voted, draining the other rank's in-flight requests first.
        hub.stop()
    dist.destroy_process_group()


def test_expert_parallel_serving_lockstep(tmp_path):
    port = 29700 + (os.getpid() + 29) % 500
    mp.spawn(_ep_serving_worker, args=(2, port, str(tmp_path)), nprocs=2,
             join=True)
