import json, os, subprocess, sys
for env, tag in ((None, "materialized"), ("1", "fused")):
    e = dict(os.environ)
    if env: e["DTX_FUSED_CE"] = env
    e["DTX_MEM_PROBE"] = "1"
    out = subprocess.run(
        [sys.executable, "-c", '''
import os, sys, torch
sys.path.insert(0, "/root/repo")
from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM
torch.manual_seed(0)
with torch.device("cuda:0"):
    m = LlamaForCausalLM(LlamaConfig.llama3_8b(), lora=True, dtype=torch.bfloat16)
m.init_random()
ids = torch.randint(3, 128256, (24, 1024), device="cuda:0")
loss = m(ids, labels=ids.clone()); loss.backward()
torch.cuda.synchronize()
print(round(torch.cuda.max_memory_allocated()/2**30, 2))
'''], env=e, capture_output=True, text=True)
    print(tag, "peak GiB:", out.stdout.strip().splitlines()[-1] if out.stdout.strip() else out.stderr[-300:])
