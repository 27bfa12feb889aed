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

# hipBLASLt/rocBLAS algorithm selections pre-tuned on MI355X (PyTorch
# TunableOp, profiles/tunableop_gfx950.csv — ~2% end-to-end on the 7B
# LoRA step). Read-only; explicit PYTORCH_TUNABLEOP_* env wins.
import os as _os

_tune = _os.path.join(_os.path.dirname(_os.path.dirname(
    _os.path.abspath(__file__))), "profiles", "tunableop_gfx950.csv")
if (_os.path.exists(_tune)
        and "PYTORCH_TUNABLEOP_ENABLED" not in _os.environ):
    _os.environ["PYTORCH_TUNABLEOP_ENABLED"] = "1"
    _os.environ["PYTORCH_TUNABLEOP_TUNING"] = "0"
    _os.environ["PYTORCH_TUNABLEOP_FILENAME"] = _tune
del _os
