"""Op dispatch: hand-written gfx950 HIP kernels on GPU, torch reference on CPU.

Policy (deliberate, see repo docs): on a GPU box the in-tree HIP extension
`_dtx_hip` is REQUIRED — any op called with CUDA(=HIP) tensors raises if the
extension failed to import, rather than silently falling back to eager
PyTorch. CPU tensors always use the fp32 torch reference implementations
(tests, plumbing configs).
"""

from __future__ import annotations

import os

import torch

from . import reference as ref

_EXT = None
_EXT_ERR: str | None = None


def _load_ext():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        from . import _dtx_hip  # built in-tree by setup.py build_ext --inplace
        _EXT = _dtx_hip
    except Exception as e:  # pragma: no cover - exercised on GPU boxes
        _EXT_ERR = f"{type(e).__name__}: {e}"
    return _EXT


def have_ext() -> bool:
    return _load_ext() is not None


def _gpu(*tensors) -> bool:
    t = tensors[0]
    if not t.is_cuda:
        return False
    ext = _load_ext()
    if ext is None:
        raise RuntimeError(
            "datatunerx_amd HIP extension (_dtx_hip) is not built but a GPU "
            f"tensor reached the op layer. Build it with `python setup.py "
            f"build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). Import error: "
            f"{_EXT_ERR}")
    return True


# --------------------------------------------------------------- RMSNorm
def rmsnorm_fwd(x, w, eps: float = 1e-5):
    if _gpu(x):
        return _EXT.rmsnorm_fwd(x, w, eps)
    return ref.rmsnorm_fwd(x, w, eps)


def rmsnorm_bwd(dy, x, w, inv, dres=None):
    """dres: optional residual-branch gradient accumulated into dx
    IN-KERNEL (the fused residual+rmsnorm backward — kills the autograd
    grad-accumulation adds on every decoder-layer residual)."""
    if _gpu(x):
        return _EXT.rmsnorm_bwd(dy, x, w, inv, dres)
    dx, dw = ref.rmsnorm_bwd(dy, x, w, inv)
    if dres is not None:
        dx = dx + dres
    return dx, dw


# ------------------------------------------------------------------ RoPE
rope_tables = ref.rope_tables


def rope_fwd(x, cos, sin, pos0: int = 0, pos_dev=None):
    if _gpu(x):
        return _EXT.rope(x, cos, sin, pos0, False, pos_dev)
    if pos_dev is not None and pos_dev.numel() > 1:
        # batched ragged decode: per-row positions
        return torch.cat([ref.rope_fwd(x[i:i + 1], cos, sin,
                                       pos0 + int(p))
                          for i, p in enumerate(pos_dev)], dim=0)
    if pos_dev is not None:
        pos0 = pos0 + int(pos_dev)
    return ref.rope_fwd(x, cos, sin, pos0)


def rope_bwd(dy, cos, sin, pos0: int = 0):
    if _gpu(dy):
        return _EXT.rope(dy, cos, sin, pos0, True)
    return ref.rope_bwd(dy, cos, sin, pos0)


# ---------------------------------------------------------------- SwiGLU
def swiglu_fwd(gate, up):
    if _gpu(gate):
        return _EXT.swiglu_fwd(gate, up)
    return ref.swiglu_fwd(gate, up)


def swiglu_bwd(dout, gate, up):
    if _gpu(gate):
        return _EXT.swiglu_bwd(dout, gate, up)
    return ref.swiglu_bwd(dout, gate, up)


# ------------------------------------------------------ cross entropy
def softmax_xent_fwd(logits, targets, ignore_index: int = -100):
    if _gpu(logits):
        return _EXT.xent_fwd(logits, targets, ignore_index)
    return ref.softmax_xent_fwd(logits, targets, ignore_index)


def softmax_xent_bwd(logits, targets, lse, dloss, ignore_index: int = -100):
    if _gpu(logits):
        return _EXT.xent_bwd(logits, targets, lse, dloss, ignore_index)
    return ref.softmax_xent_bwd(logits, targets, lse, dloss, ignore_index)


def xent_lse_merge(logits_c, targets, m_run, l_run, tgt, v0: int,
                   ignore_index: int = -100):
    if _gpu(logits_c):
        _EXT.xent_lse_merge(logits_c, targets, m_run, l_run, tgt, v0,
                            ignore_index)
        return
    ref.xent_lse_merge(logits_c, targets, m_run, l_run, tgt, v0,
                       ignore_index)


def xent_dlogits(logits_c, targets, lse, v0: int,
                 ignore_index: int = -100):
    if _gpu(logits_c):
        return _EXT.xent_dlogits(logits_c, targets, lse, v0, ignore_index)
    return ref.xent_dlogits(logits_c, targets, lse, v0, ignore_index)


# --------------------------------------------------------- attention
def attn_fwd(q, k, v, causal: bool = True, scale: float | None = None,
             len_dev=None):
    """BSHD: q [B,S,Hq,D], k/v [B,Skv,Hkv,D] -> (o [B,S,Hq,D], lse)."""
    if scale is None:
        scale = 1.0 / (q.shape[-1] ** 0.5)
    if _gpu(q):
        if q.shape[1] == 1:
            # serving decode: flash-decoding split-KV path (with S=1
            # every cached position is visible, causal or not).
            # len_dev: device-resident cache length (hipGraph decode).
            return _EXT.attn_decode(q, k, v, scale, len_dev)
        # V consumed row-major (PV B-frags via ds_read_b64_tr_b16):
        # no V pre-transpose
        return _EXT.attn_fwd(q, k, v.contiguous(), causal, scale)
    if len_dev is not None and len_dev.numel() > 1:
        # batched ragged decode on CPU: per-row cache lengths
        outs, lses = [], []
        for i in range(q.shape[0]):
            n = int(len_dev[i])
            o, ls = ref.attn_fwd(q[i:i + 1], k[i:i + 1, :n],
                                 v[i:i + 1, :n], causal, scale)
            outs.append(o)
            lses.append(ls)
        return torch.cat(outs, 0), torch.cat(lses, 0)
    if len_dev is not None:
        k = k[:, :int(len_dev)]
        v = v[:, :int(len_dev)]
    return ref.attn_fwd(q, k, v, causal, scale)


def attn_bwd(q, k, v, o, do, lse, causal: bool = True,
             scale: float | None = None):
    if scale is None:
        scale = 1.0 / (q.shape[-1] ** 0.5)
    if _gpu(q):
        return _EXT.attn_bwd(q, k, v, o, do, lse, causal, scale)
    return ref.attn_bwd(q, k, v, o, do, lse, causal, scale)


def gemv(x, w):
    """y[...,1,N] = x[...,1,K] @ w[N,K]^T — decode GEMV (single row)."""
    if _gpu(x):
        return _EXT.gemv(x, w)
    return torch.nn.functional.linear(x, w)


def gemm_nt_supported(n: int, k: int) -> bool:
    """Shapes the hand-written MFMA GEMM handles (all Llama projection /
    lm_head shapes); anything else falls back to hipBLASLt."""
    return n % 256 == 0 and k % 64 == 0 and k >= 128


def gemm_nt(a, b, src=None):
    """C[...,N] = a[...,K] @ b[N,K]^T (+ src) via the hand-written
    256x256 MFMA kernel (gemm.hip). bf16 in/out, fp32 accumulate."""
    if _gpu(a):
        return _EXT.gemm_nt(a, b, src)
    out = (a.float() @ b.float().t()).to(a.dtype)
    if src is not None:
        out = out + src.reshape(out.shape)
    return out


# -------------------------------------------------------------- LoRA
# Dropout enters the kernels one of two ways:
#   - `mask`: a materialized bf16 mask tensor (same shape as x / y)
#   - `seed` + `keep` < 1: counter-based RNG computed INSIDE the
#     kernels (splitmix64 on the element offset) — the mask is never
#     written to or read from HBM. The same (seed, offset) yields the
#     same bits in contract/wgrad/expand, so fwd and bwd agree.
def lora_contract(x, w, mask=None, seed: int = 0, keep: float = 1.0):
    if _gpu(x):
        return _EXT.lora_contract(x, w, mask, seed, keep)
    return ref.lora_contract(x, w, mask, seed, keep)


def lora_expand_add(y, t, w, scale: float, mask=None, seed: int = 0,
                    keep: float = 1.0):
    if _gpu(y):
        _EXT.lora_expand_add(y, t, w, scale, mask, seed, keep)
        return y
    return ref.lora_expand_add(y, t, w, scale, mask, seed, keep)


def lora_wgrad(t, x, scale: float = 1.0, mask=None, seed: int = 0,
               keep: float = 1.0):
    if _gpu(x):
        return _EXT.lora_wgrad(t, x, scale, mask, seed, keep)
    return ref.lora_wgrad(t, x, scale, mask, seed, keep)


def dropout_mask(M: int, K: int, seed: int, keep: float, like):
    """Materialize the RNG mask (bit-identical to the fused kernels'
    in-kernel bits) — tests and the r>16 fallback path."""
    if like.is_cuda and _EXT is not None:
        return _EXT.dropout_mask(M, K, seed, keep, like)
    return ref.dropout_mask(M, K, seed, keep, device=like.device,
                            dtype=torch.bfloat16)


# -------------------------------------------------- weight dequant
def dequant_int8(q, scale):
    if _gpu(q):
        return _EXT.dequant_int8(q, scale)
    return (q.float() * scale[:, None]).to(torch.bfloat16)


def dequant_int4(packed, scale, group: int = 64):
    if _gpu(packed):
        return _EXT.dequant_int4(packed, scale, group)
    from ..models.quant import dequantize_int4
    return dequantize_int4(packed, scale, torch.bfloat16, group)


# ------------------------------------------------------------- AdamW
def adamw_step(p_bf16, master, grad, m, v, lr, beta1, beta2, eps,
               weight_decay, step: int):
    if _gpu(master):
        _EXT.adamw(p_bf16, master, grad, m, v, lr, beta1, beta2, eps,
                   weight_decay, step)
        return
    ref.adamw_step(p_bf16, master, grad, m, v, lr, beta1, beta2, eps,
                   weight_decay, step)


def l2_norm(flat):
    if _gpu(flat):
        return _EXT.l2_norm(flat)
    return ref.l2_norm(flat)
