"""Distributed setup: one process per GPU, torch.distributed over RCCL
(backend "nccl" IS RCCL on ROCm) for xGMI collectives; gloo on CPU hosts.

The control plane needs no collectives (single host, unix sockets); this
package serves the data plane only: tensor parallelism for large backing
models and the data-parallel session-sharded serving bench.
"""
from __future__ import annotations

import os

import torch
import torch.distributed as dist

_TP_GROUP = None
_TP_RANK = 0
_TP_SIZE = 1


def init_distributed(backend: str | None = None) -> tuple[int, int]:
    """Initialize from torchrun env vars; returns (rank, world_size)."""
    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    if "RANK" not in os.environ:
        return 0, 1
    ndev = torch.cuda.device_count() if torch.cuda.is_available() else 0
    local_world = int(os.environ.get(
        "LOCAL_WORLD_SIZE", os.environ.get("WORLD_SIZE", "1")))
    if backend is None:
        # RCCL needs one DISTINCT GPU per rank; an oversubscribed smoke
        # run (more local ranks than GPUs) falls back to gloo so the
        # multi-rank harness still executes on a small box
        backend = "nccl" if 0 < local_world <= ndev else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    dist.init_process_group(backend=backend)
    rank = dist.get_rank()
    if ndev > 0:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)) % ndev)
    return rank, dist.get_world_size()


def init_tensor_parallel(tp_size: int) -> None:
    """Carve TP groups out of the world (world must be a multiple)."""
    global _TP_GROUP, _TP_RANK, _TP_SIZE
    _TP_SIZE = tp_size
    if tp_size == 1 or not dist.is_initialized():
        _TP_GROUP, _TP_RANK = None, 0
        return
    world = dist.get_world_size()
    rank = dist.get_rank()
    assert world % tp_size == 0
    for start in range(0, world, tp_size):
        ranks = list(range(start, start + tp_size))
        g = dist.new_group(ranks)
        if rank in ranks:
            _TP_GROUP = g
            _TP_RANK = rank - start
    _TP_SIZE = tp_size


def tp_rank() -> int:
    return _TP_RANK


def tp_size() -> int:
    return _TP_SIZE


def tp_group():
    return _TP_GROUP


def tp_all_reduce(t: torch.Tensor) -> torch.Tensor:
    if _TP_SIZE > 1:
        dist.all_reduce(t, group=_TP_GROUP)
    return t


# ---- comm/GEMM overlap (north star: all-reduce on a dedicated HIP
# stream overlapped with the next GEMM; used by the pipelined TP prefill
# in models/llama.py) ----
_COMM_STREAM = None


def comm_stream():
    global _COMM_STREAM
    if _COMM_STREAM is None and torch.cuda.is_available():
        _COMM_STREAM = torch.cuda.Stream()
    return _COMM_STREAM


def tp_all_reduce_async(t: torch.Tensor):
    """Launch the TP all-reduce on the comm stream; returns an event the
    consumer stream must wait on (None when already complete — CPU/gloo
    or TP=1). RCCL ties the collective to the stream current at call
    time, so compute on the default stream overlaps the transfer."""
    if _TP_SIZE <= 1:
        return None
    if not t.is_cuda:
        dist.all_reduce(t, group=_TP_GROUP)
        return None
    ready = torch.cuda.Event()
    ready.record()
    cs = comm_stream()
    with torch.cuda.stream(cs):
        cs.wait_event(ready)
        dist.all_reduce(t, group=_TP_GROUP)
        done = torch.cuda.Event()
        done.record()
    return done


def wait_comm(ev) -> None:
    if ev is not None:
        torch.cuda.current_stream().wait_event(ev)


def barrier() -> None:
    if dist.is_initialized():
        dist.barrier()


# ---- expert parallelism (MoE): experts partitioned by id across the
# world, tokens routed to their experts' owners with all-to-all over
# xGMI (RCCL) — the alternative to TP-sharding every expert's
# intermediate dim. EP spans the whole world; TP must be 1.
_EP_SIZE = 1
_EP_RANK = 0


def init_expert_parallel(ep_size: int) -> None:
    global _EP_SIZE, _EP_RANK
    assert _TP_SIZE == 1, "EP spans the world; combine with TP is not wired"
    _EP_SIZE = ep_size
    _EP_RANK = dist.get_rank() if (ep_size > 1 and dist.is_initialized()) \
        else 0


def ep_size() -> int:
    return _EP_SIZE


def ep_rank() -> int:
    return _EP_RANK


def ep_all_to_all(out: torch.Tensor, inp: torch.Tensor,
                  out_splits, in_splits) -> None:
    if out.is_cuda and dist.get_backend() == "gloo":
        # oversubscribed rehearsal (more ranks than GPUs falls back to
        # gloo, which cannot move CUDA tensors): stage through host
        h_out = torch.empty_like(out, device="cpu")
        dist.all_to_all_single(h_out, inp.cpu(), out_splits, in_splits)
        out.copy_(h_out)
        return
    dist.all_to_all_single(out, inp, out_splits, in_splits)
