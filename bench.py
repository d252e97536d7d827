#!/usr/bin/env python3
"""Flagship serving benchmark: agent turns/sec, 64 Sessions on Llama-3-8B.

Driver contract: `python bench.py --gpus N --steps K --warmup W` (for N>1 the
driver launches via torch.distributed.run, one rank per GPU over RCCL).
One *step* = one lockstep round in which EVERY session completes one agent
turn (prompt prefill + decode) against its persistent paged-KV context.
Sessions are sharded across GPUs (data parallel — the production answer for
an 8B backing model on an 8-GPU node); per-GPU work is fixed => weak scaling.

Rank 0 prints one JSON line with the whole-job aggregate turns/sec.
"""
from __future__ import annotations

import argparse
import json
import os
import statistics
import time

import torch


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--model", default="llama-3-8b")
    ap.add_argument("--sessions", type=int, default=64)
    ap.add_argument("--prompt-len", type=int, default=512)
    ap.add_argument("--followup-len", type=int, default=256)
    ap.add_argument("--decode-len", type=int, default=128)
    ap.add_argument("--ctx-cap", type=int, default=3584)
    ap.add_argument("--max-model-len", type=int, default=4096)
    ap.add_argument("--no-graphs", action="store_true")
    ap.add_argument("--decode-microbatch", type=int, default=64)
    ap.add_argument("--kv-dtype", choices=["bf16", "fp8"], default="bf16")
    ap.add_argument("--moe-ep", action="store_true",
                    help="MoE expert parallelism: experts partitioned "
                         "across ranks, tokens routed over all-to-all "
                         "(RCCL over xGMI); sessions stay rank-local")
    ap.add_argument("--device", default=None)
    args = ap.parse_args()

    from kukeon_amd import parallel
    from kukeon_amd.engine.config import (EngineConfig, MODEL_PRESETS,
                                          SamplingParams)
    from kukeon_amd.engine.engine import LLMEngine
    from kukeon_amd.serve.sessions import TurnDriver

    rank, world = parallel.init_distributed()
    n_gpus = max(args.gpus, world)

    if args.device:
        device = args.device
    elif torch.cuda.is_available():
        dev_i = int(os.environ.get("LOCAL_RANK", 0)) % torch.cuda.device_count()
        device = f"cuda:{dev_i}"
        torch.cuda.set_device(device)
    else:
        device = "cpu"

    cfg = MODEL_PRESETS[args.model]()
    tp_size = world if args.model == "llama-3-70b" else 1
    # TP ranks share a sampling seed (identical draws, no collective needed)
    ecfg = EngineConfig(max_model_len=args.max_model_len,
                        max_sessions=max(args.sessions, 8),
                        use_graphs=(not args.no_graphs and device != "cpu"),
                        decode_microbatch=args.decode_microbatch,
                        kv_dtype=args.kv_dtype,
                        seed=1234 + rank // tp_size)
    if device == "cpu":
        ecfg.num_kv_blocks = (args.sessions *
                              (args.ctx_cap // ecfg.block_size + 2) + 64)

    if args.model == "llama-3-70b":
        parallel.init_tensor_parallel(world)
        ecfg.tp_size = world
    use_ep = args.moe_ep and world > 1
    if use_ep:
        parallel.init_expert_parallel(world)
        ecfg.use_graphs = False  # all-to-all splits are data-dependent
    if cfg.is_moe:
        from kukeon_amd.models.mixtral import MixtralModel
        model = MixtralModel(cfg, device=device)
    else:
        from kukeon_amd.models.llama import LlamaModel
        model = LlamaModel(cfg, device=device)

    engine = LLMEngine(model, cfg, ecfg, device=device)
    tp = ecfg.tp_size
    dp = max(1, world // tp)
    sessions_per_engine = args.sessions
    vocab = cfg.vocab_size
    sampling = SamplingParams(temperature=0.7, top_k=50, top_p=0.9,
                              max_new_tokens=args.decode_len)
    driver = TurnDriver(engine, sessions_per_engine, vocab,
                        first_prompt=args.prompt_len,
                        followup_prompt=args.followup_len,
                        decode_len=args.decode_len, ctx_cap=args.ctx_cap,
                        sampling=sampling, seed=1234 + (rank // tp))

    def sync():
        if device != "cpu":
            torch.cuda.synchronize()
        parallel.barrier()

    engine.capture_all()
    for _ in range(args.warmup):
        driver.run_round()
    driver.turn_latencies.clear()

    sync()
    t0 = time.perf_counter()
    turns = 0
    for _ in range(args.steps):
        turns += driver.run_round()
    sync()
    elapsed = time.perf_counter() - t0

    # max over ranks (slowest rank defines the job)
    if world > 1:
        import torch.distributed as dist
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if device != "cpu" else None)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t[0])

    total_turns = turns * dp  # TP ranks serve the same sessions once
    value = total_turns / elapsed
    p50 = (statistics.median(driver.turn_latencies) * 1000
           if driver.turn_latencies else None)

    if rank == 0:
        out = {
            "metric": "agent_turns_per_sec",
            "value": round(value, 3),
            "unit": "turns/s",
            "n_gpus": n_gpus,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            # weights/compute are always bf16; a reduced-precision KV
            # cache is reported so the line never overstates the dtype
            "dtype": ("bf16" if args.kv_dtype == "bf16"
                      else f"bf16+kv_{args.kv_dtype}"),
            "data": "synthetic",
            "p50_turn_latency_ms": round(p50, 1) if p50 else None,
            "config": {
                "model": args.model,
                "global_batch": sessions_per_engine * dp,
                "seq_len": args.ctx_cap,
                "parallelism": (f"ep{world}" if use_ep else
                                f"tp{tp}" if tp > 1 else f"dp{dp}"),
                "sessions_per_gpu_group": sessions_per_engine,
                "prompt_len": args.prompt_len,
                "followup_len": args.followup_len,
                "decode_len": args.decode_len,
                "turns_per_step": sessions_per_engine * dp,
            },
        }
        print(json.dumps(out))


if __name__ == "__main__":
    main()
