"""Measure fused-projection GEMMs vs split ones on MI355X.

Question: does one [M,E]x[E,Nq+Nk+Nv] GEMM beat three separate
projections (and one [M,E]x[E,2I] beat gate+up) for the 7B training
shapes?  If the fused N-dimension lifts hipBLASLt efficiency by >=3%,
the model should fuse its frozen base projections (the LoRA low-rank
terms stay separate ops on slices).

Run: python tools/bench_fused_gemm.py  (GPU box)
"""

import torch


def t_ms(fn, iters=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    s = torch.cuda.Event(enable_timing=True)
    e = torch.cuda.Event(enable_timing=True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters


def bench_group(tag, M, E, ns):
    x = torch.randn(M, E, device="cuda", dtype=torch.bfloat16)
    ws = [torch.randn(n, E, device="cuda", dtype=torch.bfloat16)
          for n in ns]
    wf = torch.randn(sum(ns), E, device="cuda", dtype=torch.bfloat16)
    split = t_ms(lambda: [torch.nn.functional.linear(x, w) for w in ws])
    fused = t_ms(lambda: torch.nn.functional.linear(x, wf))
    fl = 2.0 * M * E * sum(ns)
    print(f"{tag}: split {split:.3f} ms ({fl/split/1e9:.0f} TF/s)  "
          f"fused {fused:.3f} ms ({fl/fused/1e9:.0f} TF/s)  "
          f"speedup {split/fused:.3f}x")
    # dgrad shapes: dy @ W  ->  [M, N] x [N, E]
    dys = [torch.randn(M, n, device="cuda", dtype=torch.bfloat16)
           for n in ns]
    dyf = torch.randn(M, sum(ns), device="cuda", dtype=torch.bfloat16)
    split = t_ms(lambda: [dy @ w for dy, w in zip(dys, ws)])
    fused = t_ms(lambda: dyf @ wf)
    print(f"{tag} dgrad: split {split:.3f} ms ({fl/split/1e9:.0f} TF/s)  "
          f"fused {fused:.3f} ms ({fl/fused/1e9:.0f} TF/s)  "
          f"speedup {split/fused:.3f}x")


if __name__ == "__main__":
    M, E = 24 * 1024, 4096          # mb24 x seq1024, 7B hidden
    print("== 7B shapes (mb24) ==")
    bench_group("qkv", M, E, [4096, 4096, 4096])
    bench_group("gate+up", M, E, [11008, 11008])
    print("== 13B shapes (mb12, E=5120) ==")
    bench_group("qkv", 12 * 1024, 5120, [5120, 5120, 5120])
    bench_group("gate+up", 12 * 1024, 5120, [13824, 13824])
    print("== llama3-8b GQA (mb24): q 4096, kv 1024 each ==")
    bench_group("qkv", M, 4096, [4096, 1024, 1024])
    bench_group("gate+up", M, 4096, [14336, 14336])
