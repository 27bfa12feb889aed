"""Serving decode throughput (single stream): prefill a prompt, then
time N decoded tokens through the hipGraph-captured token step.

Run on the GPU box: python tools/bench_decode.py [--no-graph]
"""

import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM  # noqa: E402
from datatunerx_amd.serve.engine import InferenceEngine  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--no-graph", action="store_true")
    ap.add_argument("--tokens", type=int, default=64)
    args = ap.parse_args()
    dev = torch.device("cuda")
    model = LlamaForCausalLM(LlamaConfig.llama2_7b(), lora=False,
                             dtype=torch.bfloat16)
    model.to(dev)
    model.init_random(seed=0)
    model.eval()
    eng = InferenceEngine(model, template="vanilla", device=dev,
                          graph_decode=not args.no_graph)
    msgs = [{"role": "user", "content": "tell me a story " * 24}]
    eng.chat(msgs, 8)                      # warmup (captures the graph)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    eng.chat(msgs, args.tokens)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(f"graph={not args.no_graph}: {args.tokens} tokens in "
          f"{dt*1e3:.1f} ms = {args.tokens/dt:.1f} tok/s")


if __name__ == "__main__":
    main()
