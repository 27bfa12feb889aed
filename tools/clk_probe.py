import sys, torch
sys.path.insert(0, "/root/repo")
from datatunerx_amd import ops
a = (torch.rand(24576, 4096, device="cuda") * 2 - 1).to(torch.bfloat16)
b = (torch.rand(4096, 4096, device="cuda") * 2 - 1).to(torch.bfloat16)
for _ in range(3):
    ops.gemm_nt(a, b); torch.nn.functional.linear(a, b)
torch.cuda.synchronize()
for _ in range(5):
    ops.gemm_nt(a, b)
torch.cuda.synchronize()
for _ in range(5):
    torch.nn.functional.linear(a, b)
torch.cuda.synchronize()
print("done")
