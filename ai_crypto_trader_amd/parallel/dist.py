"""torch.distributed (RCCL over xGMI) helpers — one process per GPU.

The reference's only "distributed" mechanism is Redis pub/sub between
service containers (docker-compose.yml). The MI355X-native
data plane instead shards numeric work across the 8 GPUs of a node with
RCCL collectives (backend "nccl" IS RCCL on ROCm): GA fitness all-gather,
MC stat all-reduce, DDP gradient buckets. xGMI is point-to-point (7 links
x ~153 GB/s per GPU), so small latency-bound collectives (KB fitness
vectors) prefer direct all-gather over rings — payloads here are tiny, so
a single fused all-gather per generation is the design point.
"""

from __future__ import annotations

import os
from datetime import timedelta

import torch
import torch.distributed as dist


def env_rank() -> int:
    return int(os.environ.get("RANK", "0"))


def env_world_size() -> int:
    return int(os.environ.get("WORLD_SIZE", "1"))


def env_local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))


def init_distributed(backend: str | None = None) -> tuple[int, int, torch.device]:
    """Initialize torch.distributed from torchrun-style env vars.

    Returns (rank, world_size, device). Single-process (WORLD_SIZE absent
    or 1) does not create a process group. Backend defaults to nccl(=RCCL)
    when CUDA/HIP devices are visible, else gloo.
    """
    world = env_world_size()
    rank = env_rank()
    use_cuda = torch.cuda.is_available()
    if use_cuda:
        local = env_local_rank()
        torch.cuda.set_device(local % torch.cuda.device_count())
        device = torch.device("cuda", local % torch.cuda.device_count())
    else:
        device = torch.device("cpu")
    if world > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if use_cuda else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29571")
        dist.init_process_group(
            backend=backend, rank=rank, world_size=world,
            timeout=timedelta(minutes=10),
        )
    return rank, world, device


def is_dist() -> bool:
    return dist.is_available() and dist.is_initialized()


def barrier():
    if is_dist():
        dist.barrier()


def all_gather_rows(t: torch.Tensor) -> torch.Tensor:
    """All-gather along dim 0: (n, ...) per rank -> (world*n, ...)."""
    if not is_dist():
        return t
    world = dist.get_world_size()
    out = torch.empty(
        (world * t.shape[0],) + tuple(t.shape[1:]),
        dtype=t.dtype, device=t.device,
    )
    if dist.get_backend() == "nccl":
        dist.all_gather_into_tensor(out, t.contiguous())
    else:  # gloo (CPU tests) has no all_gather_into_tensor
        chunks = list(out.chunk(world, dim=0))
        dist.all_gather(chunks, t.contiguous())
    return out


def all_reduce_sum_(t: torch.Tensor) -> torch.Tensor:
    if is_dist():
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return t


def all_reduce_max_scalar(x: float, device) -> float:
    if not is_dist():
        return x
    t = torch.tensor([x], dtype=torch.float64, device=device)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    return float(t.item())


def destroy():
    if is_dist():
        dist.destroy_process_group()
