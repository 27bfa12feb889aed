#!/usr/bin/env python3
"""rccl-tests-style collective self-check: bit-exact
correctness + bus bandwidth for the collectives the trainer/TP engine
use (all-reduce, reduce-scatter, all-gather, broadcast, all-to-all),
over RCCL/xGMI on GPUs or gloo on CPU.

  torchrun --nnodes=1 --nproc-per-node N --master-addr 127.0.0.1 \
      tools/rccl_check.py [--bytes 268435456] [--iters 20]

Prints one line per collective per rank-0: max |err| vs the analytic
expectation and algorithmic bus GB/s (ring-model factors as in
rccl-tests: allreduce 2(n-1)/n, reducescatter/allgather (n-1)/n).
"""
import argparse
import os
import sys
import time

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--bytes", type=int, default=1 << 28)
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    use_gpu = torch.cuda.is_available()
    backend = "nccl" if use_gpu else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29531")
    dist.init_process_group(backend, rank=rank, world_size=world)
    dev = torch.device(f"cuda:{int(os.environ.get('LOCAL_RANK', rank))}"
                       if use_gpu else "cpu")
    if use_gpu:
        torch.cuda.set_device(dev)
    n = max(1024, args.bytes // 4)
    dt = torch.float32

    def sync():
        if use_gpu:
            torch.cuda.synchronize()
        dist.barrier()

    def bench(fn, check, name, factor):
        fn()                                     # correctness pass
        err = check()
        sync()
        t0 = time.perf_counter()
        for _ in range(args.iters):
            fn()
        sync()
        dt_s = (time.perf_counter() - t0) / args.iters
        bus = (n * 4) * factor / dt_s / 1e9
        if rank == 0:
            print(f"{name:15s} bytes {n * 4:>12d}  err {err:.3e}  "
                  f"busbw {bus:8.2f} GB/s  {dt_s * 1e3:8.3f} ms")
        assert err == 0.0, f"{name}: nonzero error {err}"

    # deterministic integer-valued floats => collectives must be EXACT
    base = torch.arange(n, device=dev, dtype=dt) % 1024

    # all-reduce (sum): expect sum over ranks
    x = (base + rank).clone()
    bench(lambda: dist.all_reduce(x.copy_(base + rank)),
          lambda: float((x - (base * world + world * (world - 1) / 2))
                        .abs().max()),
          "all_reduce", 2.0 * (world - 1) / world)

    # reduce-scatter
    shard = torch.empty(n // world, device=dev, dtype=dt)
    ins = list((base + rank).chunk(world))

    def rs():
        if backend == "gloo":
            y = (base + rank).clone()
            dist.all_reduce(y)
            shard.copy_(y.chunk(world)[rank])
        else:
            dist.reduce_scatter_tensor(shard, base + rank)
    exp = (base * world + world * (world - 1) / 2).chunk(world)[rank]
    bench(rs, lambda: float((shard - exp).abs().max()),
          "reduce_scatter", (world - 1) / world)

    # all-gather
    full = torch.empty(n, device=dev, dtype=dt)
    mine = base.chunk(world)[rank] + rank

    def ag():
        if backend == "gloo":
            outs = list(full.chunk(world))
            dist.all_gather(outs, mine)
        else:
            dist.all_gather_into_tensor(full, mine)
    expf = torch.cat([base.chunk(world)[r] + r for r in range(world)])
    bench(ag, lambda: float((full - expf).abs().max()),
          "all_gather", (world - 1) / world)

    # broadcast
    y = base.clone() if rank == 0 else torch.zeros_like(base)
    bench(lambda: dist.broadcast(y, src=0),
          lambda: float((y - base).abs().max()), "broadcast", 1.0)

    # all-to-all (TP/EP building block; gloo lacks it)
    if backend == "nccl":
        outb = torch.empty(n, device=dev, dtype=dt)
        inb = torch.cat([base.chunk(world)[r] + rank
                         for r in range(world)])
        expa = torch.cat([base.chunk(world)[rank] + r
                          for r in range(world)])
        bench(lambda: dist.all_to_all_single(outb, inb),
              lambda: float((outb - expa).abs().max()),
              "all_to_all", (world - 1) / world)

    if rank == 0:
        print("rccl_check: ALL EXACT")
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
