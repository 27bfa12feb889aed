#!/usr/bin/env python3
"""Attention kernel microbench on the training shape.

Reports ms + effective TF/s per kernel (causal ~halves the useful
flops; we count the causal flops actually computed: full tiles below
the diagonal + masked diagonal tiles)."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from datatunerx_amd import ops
from datatunerx_amd.ops import _dtx_hip

assert torch.cuda.is_available()
dev = torch.device("cuda:0")
B, H, S, D = 16, 32, 1024, 128
scale = D ** -0.5

q = torch.randn(B, S, H, D, device=dev, dtype=torch.bfloat16)
k = torch.randn_like(q)
v = torch.randn_like(q)
do = torch.randn_like(q)


def bench(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


# causal: ~1/2 the S^2 work
fwd_fl = 4 * B * H * S * S * D * 0.5
bwd_dkdv_fl = fwd_fl / 2 * 5          # S,dP,dV,dK + exp overhead ~ 2.5x fwd
bwd_dq_fl = fwd_fl / 2 * 3

t = bench(lambda: ops.attn_fwd(q, k, v, True, scale))
print(f"attn_fwd (incl V-transpose): {t*1e3:7.3f} ms  {fwd_fl/t/1e12:6.0f} TF/s")

t = bench(lambda: _dtx_hip.attn_fwd(q, k, v, True, scale))
print(f"attn_fwd (kernel only):      {t*1e3:7.3f} ms  {fwd_fl/t/1e12:6.0f} TF/s")

o, lse = ops.attn_fwd(q, k, v, True, scale)
t = bench(lambda: _dtx_hip.attn_bwd(q, k, v, o, do, lse, True, scale))
print(f"attn_bwd (all):              {t*1e3:7.3f} ms  {(bwd_dkdv_fl+bwd_dq_fl)/t/1e12:6.0f} TF/s")

t = bench(lambda: _dtx_hip.transpose_sd(q))
gb = q.numel() * 2 * 2 / 1e9
print(f"transpose_sd:                {t*1e3:7.3f} ms  {gb/t/1e3:6.2f} TB/s")
