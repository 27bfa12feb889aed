"""Kernel-level timing for the fused LoRA ops (7B training shapes).

Run on the GPU box: python tools/bench_lora.py
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from datatunerx_amd.ops import lora_contract, lora_expand_add, lora_wgrad


def t_us(fn, iters=50):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000


def run(M, K, N, r):
    dev = "cuda"
    x = torch.randn(M, K, device=dev, dtype=torch.bfloat16)
    a = torch.randn(r, K, device=dev, dtype=torch.bfloat16)
    bt = torch.randn(N, r, device=dev, dtype=torch.bfloat16)
    mask = torch.bernoulli(torch.full((M, K), 0.9, device=dev)).to(
        torch.bfloat16)
    maskn = torch.bernoulli(torch.full((M, N), 0.9, device=dev)).to(
        torch.bfloat16)
    t = lora_contract(x, a)
    tn = torch.randn(M, r, device=dev, dtype=torch.float32)
    y = torch.randn(M, N, device=dev, dtype=torch.bfloat16)

    def gbs(nbytes, us):
        return nbytes / us / 1e3

    for tag, fn, nb in [
        ("contract      ", lambda: lora_contract(x, a), 2 * M * K),
        ("contract mask ", lambda: lora_contract(x, a, mask), 4 * M * K),
        ("expand        ", lambda: lora_expand_add(y, tn, bt, 1.0),
         4 * M * N),
        ("expand mask   ", lambda: lora_expand_add(y, tn, bt, 1.0, maskn),
         6 * M * N),
        ("wgrad         ", lambda: lora_wgrad(t, x, 1.0), 2 * M * K),
        ("wgrad mask    ", lambda: lora_wgrad(t, x, 1.0, mask), 4 * M * K),
    ]:
        us = t_us(fn)
        print(f"M{M} K{K} N{N} r{r} {tag} {us:8.1f} us  "
              f"{gbs(nb, us):6.2f} TB/s")


if __name__ == "__main__":
    run(24 * 1024, 4096, 4096, 8)        # 7B q/v proj, mb24
    run(12 * 1024, 5120, 5120, 8)        # 13B
    run(24 * 1024, 4096, 4096, 64)       # high-rank fallback path
