#!/usr/bin/env python3
"""One profiled training step with torch.profiler: attributes every GPU
kernel to the aten op and python line that launched it (finds stray
eager-PyTorch work the rocprof kernel list can't attribute)."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from datatunerx_amd.data.dataset import SFTDataset
from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM
from datatunerx_amd.train.trainer import SFTTrainer, TrainerConfig

device = torch.device("cuda:0")
cfg = LlamaConfig.llama2_7b(lora_dropout=0.1)
with torch.device(device):
    model = LlamaForCausalLM(cfg, lora=True, dtype=torch.bfloat16)
model.init_random(seed=1)
ds = SFTDataset.synthetic(64, 1024, cfg.vocab_size, seed=7)
tr = SFTTrainer(model, ds,
                TrainerConfig(output_dir="gpurun_out/trace_out",
                              max_steps=10 ** 9, micro_batch_size=16,
                              logging_steps=0, lora_dropout=0.1),
                device=device)
it = iter(tr.train_loader)
for _ in range(3):
    tr.train_step([next(it)])
torch.cuda.synchronize()

from torch.profiler import ProfilerActivity, profile

with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
             record_shapes=True) as prof:
    tr.train_step([next(it)])
    torch.cuda.synchronize()

print(prof.key_averages(group_by_input_shape=True).table(
    sort_by="self_cuda_time_total", row_limit=40, max_src_column_width=80))
