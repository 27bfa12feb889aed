"""Weight-only quantization for frozen base weights (the reference's
bitsandbytes int8/int4 path — cmd/tuning/train.py:224-234, Hyperparameter
int4/int8 flags mapped at finetune_controller.go:486-494).

MI355X-first design: with 288 GB HBM3E the capacity motivation is weak,
so this is a fidelity feature — weights are stored quantized
(int8 per-row absmax, or int4 group-64 absmax like nf4's grouping) and
dequantized to the compute dtype on the fly; the LoRA low-rank path and
all activations stay bf16. Dequant is one elementwise pass per forward
(memory-bound, ~1% of the GEMM it feeds).
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F


def quantize_int8(w: torch.Tensor):
    """Per-output-row absmax int8. Returns (q [N,K] int8, scale [N] f32)."""
    absmax = w.float().abs().amax(dim=1).clamp(min=1e-8)
    scale = absmax / 127.0
    q = torch.clamp(torch.round(w.float() / scale[:, None]),
                    -127, 127).to(torch.int8)
    return q, scale


def dequantize_int8(q: torch.Tensor, scale: torch.Tensor, dtype):
    return (q.float() * scale[:, None]).to(dtype)


def quantize_int4(w: torch.Tensor, group: int = 64):
    """Group-wise absmax int4 (values -7..7 packed two per byte)."""
    N, K = w.shape
    assert K % group == 0, "K % group"
    wf = w.float().view(N, K // group, group)
    absmax = wf.abs().amax(dim=2).clamp(min=1e-8)          # [N, K/g]
    scale = absmax / 7.0
    q = torch.clamp(torch.round(wf / scale[:, :, None]), -7, 7)
    q = (q + 8).to(torch.uint8).view(N, K)                 # 1..15 biased
    packed = (q[:, 0::2] | (q[:, 1::2] << 4)).contiguous()  # [N, K/2]
    return packed, scale


def dequantize_int4(packed: torch.Tensor, scale: torch.Tensor, dtype,
                    group: int = 64):
    N, K2 = packed.shape
    K = K2 * 2
    lo = (packed & 0xF).to(torch.int16) - 8
    hi = (packed >> 4).to(torch.int16) - 8
    q = torch.stack([lo, hi], dim=2).view(N, K).float()
    q = q.view(N, K // group, group) * scale[:, :, None].float()
    return q.view(N, K).to(dtype)


class QuantFrozenLinear(nn.Module):
    """Drop-in for FrozenLinear holding int8/int4 weights; dequantizes
    per forward. `weight` property materializes bf16 on demand so LoRA
    wrappers and checkpoint code keep working."""

    def __init__(self, in_features: int, out_features: int,
                 bits: int = 8, dtype=torch.bfloat16, group: int = 64):
        super().__init__()
        assert bits in (4, 8)
        self.in_features, self.out_features = in_features, out_features
        self.bits, self.group, self.dtype = bits, group, dtype
        if bits == 8:
            self.register_buffer(
                "qweight", torch.zeros(out_features, in_features,
                                       dtype=torch.int8))
            self.register_buffer(
                "scale", torch.ones(out_features, dtype=torch.float32))
        else:
            self.register_buffer(
                "qweight", torch.zeros(out_features, in_features // 2,
                                       dtype=torch.uint8))
            self.register_buffer(
                "scale", torch.ones(out_features, in_features // group,
                                    dtype=torch.float32))

    @classmethod
    def from_weight(cls, w: torch.Tensor, bits: int = 8, group: int = 64):
        with torch.device(w.device):
            m = cls(w.shape[1], w.shape[0], bits=bits, dtype=w.dtype,
                    group=group)
        m.load_weight(w)
        return m

    @torch.no_grad()
    def load_weight(self, w: torch.Tensor):
        if self.bits == 8:
            q, s = quantize_int8(w)
        else:
            q, s = quantize_int4(w, self.group)
        self.qweight.copy_(q)
        self.scale.copy_(s)

    @property
    def weight(self):
        if self.qweight.is_cuda and self.dtype == torch.bfloat16:
            from ..ops import dequant_int4, dequant_int8
            if self.bits == 8:
                return dequant_int8(self.qweight, self.scale)
            return dequant_int4(self.qweight, self.scale, self.group)
        if self.bits == 8:
            return dequantize_int8(self.qweight, self.scale, self.dtype)
        return dequantize_int4(self.qweight, self.scale, self.dtype,
                               self.group)

    def forward(self, x):
        return F.linear(x, self.weight)


@torch.no_grad()
def quantize_model_(model: nn.Module, bits: int = 8):
    """Replace every FrozenLinear (and the frozen base weight inside
    LoRA modules stays bf16 — PEFT quantizes only the base model's
    nn.Linear layers; our LoRA modules fuse base+adapter, so they are
    left at bf16 exactly like lm_head fp32-forcing in the reference).
    Returns the count of quantized layers."""
    from .lora import FrozenLinear
    n = 0
    for parent in model.modules():
        for name, child in list(parent.named_children()):
            if isinstance(child, FrozenLinear):
                q = QuantFrozenLinear.from_weight(
                    child.weight.detach(), bits=bits)
                setattr(parent, name, q)
                n += 1
    return n
