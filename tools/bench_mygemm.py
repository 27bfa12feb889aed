#!/usr/bin/env python3
"""A/B the hand-written MFMA GEMM (ops/hip/gemm.hip) against hipBLASLt on
the mb24 training shapes (VERDICT round-1 item 1). Within-probe
interleaved rounds (guide §5.4 rule 24); random [-1,1) data (rule 25).

  gpurun -- 'python tools/bench_mygemm.py'
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
from datatunerx_amd import ops  # noqa: E402

assert torch.cuda.is_available() and ops.have_ext()
dev = torch.device("cuda:0")
torch.manual_seed(0)

M = 24576
shapes = [
    ("qkvo  ", M, 4096, 4096),
    ("gateup", M, 11008, 4096),
    ("down  ", M, 4096, 11008),
    ("lmhead", M, 32000, 4096),
]


def timeit(fn, iters):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


for name, m, n, k in shapes:
    a = (torch.rand((m, k), device=dev) * 2 - 1).to(torch.bfloat16)
    b = (torch.rand((n, k), device=dev) * 2 - 1).to(torch.bfloat16)
    src = (torch.rand((m, n), device=dev) * 2 - 1).to(torch.bfloat16)

    # correctness first (fp32 torch reference of the bf16 inputs)
    c = ops.gemm_nt(a, b)
    with torch.no_grad():
        refc = (a[: 512].float() @ b.float().t())
    err = (c[: 512].float() - refc).abs().max() / refc.abs().max()
    cs = ops.gemm_nt(a, b, src)
    errs = (cs[: 512].float() - (refc + src[: 512].float())).abs().max() \
        / refc.abs().max()
    print(f"{name} M={m} N={n} K={k}  rel-err {err:.4e}  src-err {errs:.4e}")
    del c, cs

    fl = 2.0 * m * n * k
    # warmup both
    for _ in range(3):
        ops.gemm_nt(a, b)
        torch.nn.functional.linear(a, b)
    # interleaved A/B rounds
    mine, lib = [], []
    for _ in range(5):
        mine.append(timeit(lambda: ops.gemm_nt(a, b), 5))
        lib.append(timeit(lambda: torch.nn.functional.linear(a, b), 5))
    tm, tl = min(mine), min(lib)
    print(f"   mine {tm*1e3:7.3f} ms ({fl/tm/1e12:7.1f} TF/s)   "
          f"hipblaslt {tl*1e3:7.3f} ms ({fl/tl/1e12:7.1f} TF/s)   "
          f"ratio {tl/tm:5.2f}x")
    del a, b, src
    torch.cuda.empty_cache()


# ---- dgrad comparison: lib dy@W (NN layout) vs mine dy@(W^T)^T (NT via
# the cached transpose). The W^T trick means our dgrad runs at NT speed.
print("\n==== dgrad ====")
for name, m, n, k in shapes:
    dy = (torch.rand((m, n), device=dev) * 2 - 1).to(torch.bfloat16)
    w = (torch.rand((n, k), device=dev) * 2 - 1).to(torch.bfloat16)
    wt = w.t().contiguous()
    ref = dy[:256].float() @ w.float()
    got = ops.gemm_nt(dy, wt)
    err = (got[:256].float() - ref).abs().max() / ref.abs().max()
    fl = 2.0 * m * n * k
    for _ in range(3):
        ops.gemm_nt(dy, wt)
        dy @ w
    mine, lib = [], []
    for _ in range(5):
        mine.append(timeit(lambda: ops.gemm_nt(dy, wt), 5))
        lib.append(timeit(lambda: dy @ w, 5))
    tm, tl = min(mine), min(lib)
    print(f"{name} dgrad err {err:.2e}  mine {tm*1e3:7.3f} ms "
          f"({fl/tm/1e12:7.1f} TF/s)  lib {tl*1e3:7.3f} ms "
          f"({fl/tl/1e12:7.1f} TF/s)  ratio {tl/tm:5.2f}x")
    del dy, w, wt
    torch.cuda.empty_cache()
