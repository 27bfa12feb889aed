"""LoRA modules + HF-PEFT-compatible adapter checkpoint layout.

Adapter layout contract (what the reference produces via PEFT
`trainer.save_model` — cmd/tuning/train.py:300, SURVEY.md §5
Checkpoint/resume): a directory with `adapter_config.json` and
`adapter_model.safetensors` whose keys are
`base_model.model.<module_path>.lora_A.weight` ([r, in]) and
`...lora_B.weight` ([out, r]) — the exact names/shapes PEFT writes
(layout-verified by tests; peft itself is not in this offline image).
"""

from __future__ import annotations

import json
import math
import os

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import gemm_nt_supported, lora_contract, lora_expand_add
from ..ops.autograd import FrozenGemm
from ..ops.autograd import LoRALinear as _FusedLoRAFn


def _weight_t(mod: nn.Module):
    """Cached W^T for a frozen base weight: both the forward (x @ W^T)
    and the dgrad (dy @ W) then run the hand-written NT MFMA GEMM with
    the contraction contiguous — 288 GB HBM3E per MI355X makes the
    persistent transposed copy free. Built lazily on first GPU forward;
    None when the shape is outside the kernel's support.

    Gated by DTX_CUSTOM_GEMM=1: the hand-written kernel currently
    measures 1.25-1.33 PF/s on the mb24 training shapes vs the tuned
    hipBLASLt picks' 1.47-1.67 (tools/bench_mygemm.py, within-probe
    A/B), so routing the 60%-of-step base GEMMs through it would
    regress tokens/s; it stays opt-in until it wins."""
    if os.environ.get("DTX_CUSTOM_GEMM", "0") != "1":
        return None
    wt = getattr(mod, "_wt", False)
    if wt is False:
        w = mod.weight
        n, k = w.shape
        if (w.is_cuda and w.dtype == torch.bfloat16
                and gemm_nt_supported(n, k) and gemm_nt_supported(k, n)):
            wt = w.detach().t().contiguous()
        else:
            wt = None
        mod._wt = wt
    return wt


class FrozenLinear(nn.Module):
    """Bias-free frozen linear (Llama projections are bias-free)."""

    def __init__(self, in_features: int, out_features: int,
                 dtype=torch.bfloat16):
        super().__init__()
        self.in_features, self.out_features = in_features, out_features
        self.weight = nn.Parameter(
            torch.empty(out_features, in_features, dtype=dtype),
            requires_grad=False)

    def forward(self, x, residual=None):
        if x.is_cuda and x.numel() == x.shape[-1]:
            from ..ops import gemv
            y = gemv(x, self.weight)         # decode: streaming GEMV
            return y if residual is None else y + residual
        if x.is_cuda:
            wt = _weight_t(self)
            if wt is not None:
                # hand-written MFMA GEMM; residual add fused into the
                # kernel epilogue (no separate elementwise pass)
                return FrozenGemm.apply(x, self.weight, wt, residual)
        if residual is not None:
            # fold the residual add into the GEMM epilogue (addmm,
            # beta=1): one kernel instead of linear + elementwise add
            xs = x.shape
            x2 = x.reshape(-1, xs[-1])
            r2 = residual.reshape(-1, self.out_features)
            return torch.addmm(r2, x2, self.weight.t()).reshape(
                *xs[:-1], self.out_features)
        # hipBLASLt fallback for unsupported shapes / CPU
        return F.linear(x, self.weight)


class LoRAFunctionWithDropout(torch.autograd.Function):
    """Fused LoRA linear with PEFT-style input dropout on the low-rank
    path. The dropout mask is applied INSIDE the contract/wgrad/expand
    kernels; with `seed`-mode (r <= 16, the default ranks) it is
    RE-GENERATED from a counter-based RNG in each kernel and never
    touches HBM at all — no bernoulli launch, no mask reads. `mask` is
    the materialized-tensor fallback (r > 16 / CPU-provided masks).
    """

    @staticmethod
    def forward(ctx, x, w, a, b, scale, mask, seed, keep, wt=None):
        from ..ops.autograd import _base_gemm
        xs = x.shape
        x2 = x.reshape(-1, xs[-1])
        y = _base_gemm(x2, w, wt)
        t = lora_contract(x2, a, mask, seed, keep)   # (x o mask) @ A^T
        lora_expand_add(y, t, b, scale)
        ctx.save_for_backward(x2, w, a, b, t, mask, wt)
        ctx.scale, ctx.xshape = scale, xs
        ctx.seed, ctx.keep = seed, keep
        return y.reshape(*xs[:-1], w.shape[0])

    @staticmethod
    def backward(ctx, dy):
        from ..ops import gemm_nt, lora_wgrad
        x2, w, a, b, t, mask, wt = ctx.saved_tensors
        s, seed, keep = ctx.scale, ctx.seed, ctx.keep
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx = gemm_nt(dy2, wt) if (wt is not None and dy2.is_cuda) \
            else dy2 @ w
        dt = lora_contract(dy2, b.t().contiguous())       # [M,r] = dy @ B
        da = lora_wgrad(dt, x2, s, mask, seed, keep)  # dt^T @ (x o mask)
        db = lora_wgrad(t, dy2, s).t().contiguous()
        # dx += mask o (s * dt @ A)
        lora_expand_add(dx, dt, a.t().contiguous(), s, mask, seed, keep)
        return (dx.reshape(ctx.xshape), None, da.to(a.dtype),
                db.to(b.dtype), None, None, None, None, None)


class LoRALinearModule(nn.Module):
    """Frozen base weight + trainable LoRA A/B (HF layouts: A [r,in],
    B [out,r]); scale = alpha / r (reference defaults r=8 alpha=32 —
    cmd/tuning/parser.py:138-149)."""

    def __init__(self, in_features: int, out_features: int, r: int = 8,
                 alpha: float = 32.0, dropout: float = 0.0,
                 dtype=torch.bfloat16):
        super().__init__()
        self.in_features, self.out_features = in_features, out_features
        self.r, self.alpha, self.dropout = r, alpha, dropout
        self.scale = alpha / r
        self.weight = nn.Parameter(
            torch.empty(out_features, in_features, dtype=dtype),
            requires_grad=False)
        self.lora_A = nn.Parameter(torch.empty(r, in_features, dtype=dtype))
        self.lora_B = nn.Parameter(torch.zeros(out_features, r, dtype=dtype))
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))

    def forward(self, x):
        wt = _weight_t(self) if x.is_cuda else None
        if not (self.training and self.dropout > 0.0):
            return _FusedLoRAFn.apply(x, self.weight, self.lora_A,
                                      self.lora_B, self.scale, wt)
        keep = 1.0 - self.dropout
        # draw the per-call RNG seed from torch's generator so dropout
        # stays reproducible under torch.manual_seed
        seed = int(torch.randint(0, 2**31 - 1, (1,)).item())
        if self.r <= 16 and self.in_features % 8 == 0:
            # fused counter-based RNG: mask never materialized
            return LoRAFunctionWithDropout.apply(
                x, self.weight, self.lora_A, self.lora_B, self.scale,
                None, seed, keep, wt)
        from ..ops import dropout_mask
        mask = dropout_mask(x.numel() // x.shape[-1], x.shape[-1], seed,
                            keep, x).to(x.dtype)
        return LoRAFunctionWithDropout.apply(x, self.weight, self.lora_A,
                                             self.lora_B, self.scale,
                                             mask, 0, 1.0, wt)

    def merged_weight(self):
        """W + s·B@A — used by the serving engine (no adapter overhead)."""
        return (self.weight.float()
                + self.scale * (self.lora_B.float() @ self.lora_A.float())
                ).to(self.weight.dtype)


def lora_state_dict(model: nn.Module, prefix: str = "base_model.model."):
    out = {}
    for name, mod in model.named_modules():
        if isinstance(mod, LoRALinearModule):
            out[f"{prefix}{name}.lora_A.weight"] = mod.lora_A.detach().cpu()
            out[f"{prefix}{name}.lora_B.weight"] = mod.lora_B.detach().cpu()
    return out


def save_adapter(model: nn.Module, out_dir: str, *, r: int, alpha: float,
                 dropout: float, target_modules: list[str],
                 base_model_name_or_path: str = ""):
    """Write HF-PEFT adapter layout (adapter_config.json +
    adapter_model.safetensors)."""
    os.makedirs(out_dir, exist_ok=True)
    from safetensors.torch import save_file
    sd = {k: v.contiguous() for k, v in lora_state_dict(model).items()}
    save_file(sd, os.path.join(out_dir, "adapter_model.safetensors"))
    cfg = {
        "peft_type": "LORA",
        "task_type": "CAUSAL_LM",
        "r": r,
        "lora_alpha": alpha,
        "lora_dropout": dropout,
        "target_modules": target_modules,
        "base_model_name_or_path": base_model_name_or_path,
        "bias": "none",
        "fan_in_fan_out": False,
        "inference_mode": False,
        "modules_to_save": None,
    }
    with open(os.path.join(out_dir, "adapter_config.json"), "w") as f:
        json.dump(cfg, f, indent=2)


def load_adapter(model: nn.Module, adapter_dir: str,
                 prefix: str = "base_model.model."):
    from safetensors.torch import load_file
    sd = load_file(os.path.join(adapter_dir, "adapter_model.safetensors"))
    mods = {n: m for n, m in model.named_modules()
            if isinstance(m, LoRALinearModule)}
    n_loaded = 0
    for key, tensor in sd.items():
        if not key.startswith(prefix):
            continue
        rest = key[len(prefix):]
        for suffix, attr in ((".lora_A.weight", "lora_A"),
                             (".lora_B.weight", "lora_B")):
            if rest.endswith(suffix):
                mod_name = rest[: -len(suffix)]
                if mod_name in mods:
                    getattr(mods[mod_name], attr).data.copy_(
                        tensor.to(getattr(mods[mod_name], attr).dtype))
                    n_loaded += 1
    return n_loaded
