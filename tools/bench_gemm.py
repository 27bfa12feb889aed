#!/usr/bin/env python3
"""Time the trainer's base-GEMM shapes under the current BLAS backend.

Run twice on the GPU box to compare rocBLAS vs hipBLASLt:
  TORCH_BLAS_PREFER_HIPBLASLT=0 python tools/bench_gemm.py
  TORCH_BLAS_PREFER_HIPBLASLT=1 python tools/bench_gemm.py
"""
import os
import time

import torch

assert torch.cuda.is_available()
dev = torch.device("cuda:0")

M = 16384
shapes = [
    ("qkvo  NT", (M, 4096), (4096, 4096)),
    ("gateup NT", (M, 4096), (11008, 4096)),
    ("down  NT", (M, 11008), (4096, 11008)),
    ("lmhead NT", (M, 4096), (32000, 4096)),
]


def bench(fn, iters=30):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


print(f"TORCH_BLAS_PREFER_HIPBLASLT={os.environ.get('TORCH_BLAS_PREFER_HIPBLASLT', '<unset>')}")
total_nt = total_nn = 0.0
for name, xs, ws in shapes:
    x = torch.randn(xs, device=dev, dtype=torch.bfloat16)
    w = torch.randn(ws, device=dev, dtype=torch.bfloat16)
    dy = torch.randn((xs[0], ws[0]), device=dev, dtype=torch.bfloat16)
    t_nt = bench(lambda: x @ w.t())           # forward
    t_nn = bench(lambda: dy @ w)              # dgrad
    t_lin = bench(lambda: torch.nn.functional.linear(x, w))
    fl = 2 * xs[0] * ws[0] * ws[1]
    print(f"{name}: fwd {t_nt*1e3:7.3f} ms ({fl/t_nt/1e12:6.1f} TF/s)  "
          f"dgrad {t_nn*1e3:7.3f} ms ({fl/t_nn/1e12:6.1f} TF/s)  "
          f"F.linear {t_lin*1e3:7.3f} ms ({fl/t_lin/1e12:6.1f} TF/s)")
    total_nt += t_nt
    total_nn += t_nn
print(f"sum fwd {total_nt*1e3:.3f} ms  dgrad {total_nn*1e3:.3f} ms")
