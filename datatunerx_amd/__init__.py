"""datatunerx_amd — MI355X-native LLM fine-tuning platform.

A from-scratch rebuild of the capabilities of DataTunerX
(reference: /root/reference, a Go/KubeRay operator + Ray/HF-Trainer
payload) as an MI355X-first stack:

- declarative API objects (FinetuneExperiment / FinetuneJob / Finetune,
  LLM / Hyperparameter / Dataset / LLMCheckpoint / Scoring) with the same
  schema and state machines as the reference CRDs
  (reference: internal/controller/finetune/*.go),
- a gang scheduler that packs an experiment's concurrent jobs onto the
  8 GPUs of one MI355X node (replaces KubeRay dispatch),
- a native PyTorch-ROCm SFT/LoRA trainer whose hot path is hand-written
  HIP for CDNA4 (gfx950): fused LoRA contract/expand, flash-attention
  fwd/bwd, RMSNorm, RoPE, SwiGLU, fused cross-entropy, fused AdamW,
- RCCL-over-xGMI data parallelism (flat-bucket bf16 all-reduce) and a
  tensor-parallel inference-compare service.
"""

__version__ = "0.1.0"
