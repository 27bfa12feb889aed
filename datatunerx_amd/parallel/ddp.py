"""Data parallelism over RCCL/xGMI: one process per MI355X GPU.

Replaces the reference's Ray Train + DeepSpeed layer.
Design for the xGMI topology: each MI355X has 7 point-to-point links
(≈153 GB/s each), so LoRA's ~8 MB gradient is latency-bound — ONE fused
flat all-reduce per optimizer step; the full-param path uses fixed-size
bf16 buckets all-reduced as backward produces them (overlap), sized so a
ring transfer per bucket stays well above latency (≥64 MiB).

torch.distributed backend "nccl" IS RCCL on ROCm.
"""

from __future__ import annotations

import datetime
import os

import torch
import torch.distributed as dist


def init_distributed(backend: str | None = None):
    """Reads torchrun env (RANK/WORLD_SIZE/LOCAL_RANK/MASTER_*). Returns
    (rank, world_size, local_rank, device). Safe to call in a single
    process with no env (world_size=1, no process group)."""
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    use_gpu = torch.cuda.is_available()
    device = torch.device(f"cuda:{local_rank}" if use_gpu else "cpu")
    if use_gpu:
        torch.cuda.set_device(device)
    if world > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if use_gpu else "gloo"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        dist.init_process_group(backend=backend, rank=rank, world_size=world,
                                timeout=datetime.timedelta(seconds=600))
    return rank, world, local_rank, device


def is_main() -> bool:
    return (not dist.is_initialized()) or dist.get_rank() == 0


def sync_scalar_mean(value: float, device) -> float:
    if not dist.is_initialized():
        return value
    t = torch.tensor([value], dtype=torch.float64, device=device)
    dist.all_reduce(t, op=dist.ReduceOp.AVG)
    return float(t.item())


def barrier():
    if dist.is_initialized():
        dist.barrier()


class GradSynchronizer:
    """Gradient all-reduce strategies.

    - flat: one fused all-reduce on the whole flat grad (LoRA: ~4.2M
      params — latency-bound on xGMI, a single call per step is optimal).
    - bucketed: chunked async all-reduces (full-param models; callers may
      invoke reduce_bucket as buckets fill during backward to overlap).
    """

    def __init__(self, world_size: int, bucket_bytes: int = 64 << 20):
        self.world = world_size
        self.bucket_bytes = bucket_bytes

    def allreduce_flat_(self, flat: torch.Tensor):
        if self.world <= 1 or not dist.is_initialized():
            return
        dist.all_reduce(flat, op=dist.ReduceOp.AVG)

    def allreduce_chunked_(self, flat: torch.Tensor):
        """Bandwidth path: split into buckets, issue async all-reduces so
        RCCL pipelines them across the xGMI links."""
        if self.world <= 1 or not dist.is_initialized():
            return
        elems = max(1, self.bucket_bytes // flat.element_size())
        handles = []
        for off in range(0, flat.numel(), elems):
            chunk = flat.narrow(0, off, min(elems, flat.numel() - off))
            handles.append(dist.all_reduce(chunk, op=dist.ReduceOp.AVG,
                                           async_op=True))
        for h in handles:
            h.wait()
