"""torch.autograd.Function wrappers over the fused ops.

These are what the model code calls; each routes to the HIP kernel on GPU
and the torch reference on CPU via datatunerx_amd.ops dispatch.
"""

from __future__ import annotations

import os

import torch

from . import (attn_bwd, attn_fwd, lora_contract, lora_expand_add,
               lora_wgrad, rmsnorm_bwd, rmsnorm_fwd, rope_bwd, rope_fwd,
               softmax_xent_bwd, softmax_xent_fwd, swiglu_bwd, swiglu_fwd)


class RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, eps):
        y, inv = rmsnorm_fwd(x, w, eps)
        ctx.save_for_backward(x, w, inv)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, w, inv = ctx.saved_tensors
        dx, dw = rmsnorm_bwd(dy.contiguous(), x, w, inv)
        return dx, dw.to(w.dtype), None


def rmsnorm(x, w, eps: float = 1e-5):
    return RMSNorm.apply(x, w, eps)


class ResidualTapRMSNorm(torch.autograd.Function):
    """(norm(x), x) in one node: the layer uses output 1 as the residual
    branch, so x has ONE autograd consumer and the two branch gradients
    merge inside the rmsnorm_bwd kernel instead of a separate
    grad-accumulation add per residual (round-1 STATUS note: ~128 such
    adds/step)."""

    @staticmethod
    def forward(ctx, x, w, eps):
        y, inv = rmsnorm_fwd(x, w, eps)
        ctx.save_for_backward(x, w, inv)
        return y, x

    @staticmethod
    def backward(ctx, dy, dres):
        x, w, inv = ctx.saved_tensors
        from . import rmsnorm_bwd as _bwd
        dx, dw = _bwd(dy.contiguous(), x, w, inv,
                      dres.contiguous() if dres is not None else None)
        return dx, dw.to(w.dtype), None


def rmsnorm_tap(x, w, eps: float = 1e-5):
    return ResidualTapRMSNorm.apply(x, w, eps)


class Rope(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos, sin, pos0, pos_dev):
        ctx.save_for_backward(cos, sin)
        ctx.pos0 = pos0
        return rope_fwd(x, cos, sin, pos0, pos_dev)

    @staticmethod
    def backward(ctx, dy):
        cos, sin = ctx.saved_tensors
        return (rope_bwd(dy.contiguous(), cos, sin, ctx.pos0),
                None, None, None, None)


def rope(x, cos, sin, pos0: int = 0, pos_dev=None):
    return Rope.apply(x, cos, sin, pos0, pos_dev)


class SwiGLU(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate, up):
        ctx.save_for_backward(gate, up)
        return swiglu_fwd(gate, up)

    @staticmethod
    def backward(ctx, dout):
        gate, up = ctx.saved_tensors
        dgate, dup = swiglu_bwd(dout.contiguous(), gate, up)
        return dgate, dup


def swiglu(gate, up):
    return SwiGLU.apply(gate, up)


class Attention(torch.autograd.Function):
    """BSHD flash attention (no transpose copies around the kernel)."""

    @staticmethod
    def forward(ctx, q, k, v, causal, scale, len_dev):
        q = q.contiguous()
        if not (q.is_cuda and q.shape[1] == 1):
            # decode (S=1) accepts dense KV-cache prefix views; every
            # other path wants contiguous k/v
            k, v = k.contiguous(), v.contiguous()
        o, lse = attn_fwd(q, k, v, causal, scale, len_dev)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal, ctx.scale = causal, scale
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = attn_bwd(q, k, v, o, do.contiguous(), lse,
                              ctx.causal, ctx.scale)
        return dq, dk, dv, None, None, None


def attention(q, k, v, causal: bool = True, scale: float | None = None,
              len_dev=None):
    """q [B,S,Hq,D], k/v [B,Skv,Hkv,D] -> o [B,S,Hq,D]."""
    if scale is None:
        scale = 1.0 / (q.shape[-1] ** 0.5)
    return Attention.apply(q, k, v, causal, scale, len_dev)


class CrossEntropy(torch.autograd.Function):
    """Mean CE over non-ignored targets; logits [N,V], targets [N]."""

    @staticmethod
    def forward(ctx, logits, targets, ignore_index):
        loss, lse = softmax_xent_fwd(logits, targets, ignore_index)
        n_valid = (targets != ignore_index).sum().clamp(min=1)
        ctx.save_for_backward(logits, targets, lse, n_valid)
        ctx.ignore_index = ignore_index
        return loss.sum() / n_valid.float()

    @staticmethod
    def backward(ctx, dloss):
        logits, targets, lse, n_valid = ctx.saved_tensors
        scale = (dloss.float() / n_valid.float()).expand(logits.shape[0])
        dlogits = softmax_xent_bwd(logits, targets, lse, scale.contiguous(),
                                   ctx.ignore_index)
        return dlogits, None, None


def cross_entropy(logits, targets, ignore_index: int = -100):
    return CrossEntropy.apply(logits, targets, ignore_index)


class PerRowCrossEntropy(torch.autograd.Function):
    """UNREDUCED CE: returns the per-row NLL vector [N] (0 on ignored
    rows) through the same fused xent kernels — the building block for
    per-sequence log-probabilities (DPO)."""

    @staticmethod
    def forward(ctx, logits, targets, ignore_index):
        loss, lse = softmax_xent_fwd(logits, targets, ignore_index)
        ctx.save_for_backward(logits, targets, lse)
        ctx.ignore_index = ignore_index
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, targets, lse = ctx.saved_tensors
        dlogits = softmax_xent_bwd(logits, targets, lse,
                                   dloss.float().contiguous(),
                                   ctx.ignore_index)
        return dlogits, None, None


def per_row_cross_entropy(logits, targets, ignore_index: int = -100):
    return PerRowCrossEntropy.apply(logits, targets, ignore_index)


def _base_gemm(x2, w, wt):
    """Base-projection GEMM: the hand-written MFMA kernel (gemm.hip) when
    a cached W^T is provided (both fwd and dgrad then run the fast NT
    layout), hipBLASLt otherwise."""
    from . import gemm_nt
    if wt is not None and x2.is_cuda and x2.shape[0] > 0:
        return gemm_nt(x2, w)
    return torch.nn.functional.linear(x2, w)


class LoRALinear(torch.autograd.Function):
    """y = x @ W^T + scale * (x @ A^T) @ B^T  with frozen W.

    W: [N,K] (frozen base, no wgrad) — the base GEMM is the hand-written
    gfx950 MFMA kernel (ops/hip/gemm.hip) with `wt` = W^T cached so the
    dgrad dx = dy @ W is the same contraction-contiguous NT kernel; the
    low-rank path is the fused HIP contract/expand pair. A: [r,K],
    B: [N,r] (HF PEFT adapter layout, the checkpoint-compat contract —
    SURVEY.md §5 Checkpoint/resume).
    """

    @staticmethod
    def forward(ctx, x, w, a, b, scale, wt=None):
        xs = x.shape
        x2 = x.reshape(-1, xs[-1])
        if x2.is_cuda and x2.shape[0] == 1:
            from . import gemv
            y = gemv(x2, w)                      # decode: streaming GEMV
        else:
            y = _base_gemm(x2, w, wt)
        t = lora_contract(x2, a)                 # [M,r] f32
        lora_expand_add(y, t, b, scale)          # y += s * t @ B^T
        ctx.save_for_backward(x2, w, a, b, t, wt)
        ctx.scale = scale
        ctx.xshape = xs
        return y.reshape(*xs[:-1], w.shape[0])

    @staticmethod
    def backward(ctx, dy):
        x2, w, a, b, t, wt = ctx.saved_tensors
        s = ctx.scale
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        if wt is not None and dy2.is_cuda:
            from . import gemm_nt
            dx = gemm_nt(dy2, wt)                # base dgrad (NT MFMA)
        else:
            dx = dy2 @ w
        dt = lora_contract(dy2, b.t().contiguous())   # [M,r] = dy @ B
        db = lora_wgrad(t, dy2, s)                    # [r,N] -> B grad is [N,r]
        da = lora_wgrad(dt, x2, s)                    # [r,K]
        lora_expand_add(dx, dt, a.t().contiguous(), s)  # dx += s * dt @ A
        return (dx.reshape(ctx.xshape), None, da.to(a.dtype),
                db.t().contiguous().to(b.dtype), None, None)


def lora_linear(x, w, a, b, scale: float, wt=None):
    return LoRALinear.apply(x, w, a, b, scale, wt)


class FrozenGemm(torch.autograd.Function):
    """y = x @ W^T (+ residual) with frozen W via the MFMA NT kernel; the
    residual add runs in the GEMM epilogue (one kernel, no extra HBM
    round-trip). dgrad reuses the kernel with the cached W^T."""

    @staticmethod
    def forward(ctx, x, w, wt, residual):
        from . import gemm_nt
        xs = x.shape
        x2 = x.reshape(-1, xs[-1])
        r2 = residual.reshape(-1, w.shape[0]).contiguous() \
            if residual is not None else None
        y = gemm_nt(x2, w, r2)
        ctx.save_for_backward(wt)
        ctx.xshape, ctx.has_res = xs, residual is not None
        return y.reshape(*xs[:-1], w.shape[0])

    @staticmethod
    def backward(ctx, dy):
        from . import gemm_nt
        (wt,) = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx = gemm_nt(dy2, wt)
        return (dx.reshape(ctx.xshape), None, None,
                dy if ctx.has_res else None)


class FusedLinearCrossEntropy(torch.autograd.Function):
    """lm_head GEMM fused with cross-entropy by vocab chunking: the
    [M, V] logits tensor is NEVER materialized (SURVEY.md §2.4 row 1:
    "lm_head fused with cross-entropy to skip the logit
    materialization"). Each chunk's logits [M, Vc] stay cache-resident
    between the chunk GEMM and the online-logsumexp / dlogits kernels.

    The mean-CE backward scale (grad_out / n_valid) is a per-call
    SCALAR, so dX (and dW when the head is trainable) are accumulated
    unscaled during a second forward sweep and backward is a single
    multiply — total GEMM work equals the unfused fwd+dgrad pair while
    logits never round-trip HBM.
    """

    CHUNK = int(os.environ.get("DTX_CE_CHUNK", "4096"))
    MBLOCK = int(os.environ.get("DTX_CE_MBLOCK", "8192"))

    @staticmethod
    def forward(ctx, h2, w, targets, ignore_index, need_grads):
        from . import xent_dlogits, xent_lse_merge
        M, V = h2.shape[0], w.shape[0]
        dev = h2.device
        m = torch.full((M,), -3.4e38, device=dev, dtype=torch.float32)
        l = torch.zeros(M, device=dev, dtype=torch.float32)
        tgt = torch.zeros(M, device=dev, dtype=torch.float32)
        ch = FusedLinearCrossEntropy.CHUNK
        mb = FusedLinearCrossEntropy.MBLOCK
        # block over rows TOO: at mb=8192, ch=4096 the chunk logits
        # (64 MB), the dx row-block and the W chunk stay L3-resident
        # across the vocab sweep; without the row block a chunk is
        # ~200 MB and every re-read/accumulate round-trips HBM.
        for r0 in range(0, M, mb):
            hr = h2[r0:r0 + mb]
            tr = targets[r0:r0 + mb]
            for v0 in range(0, V, ch):
                lc = torch.nn.functional.linear(hr, w[v0:v0 + ch])
                xent_lse_merge(lc, tr, m[r0:r0 + mb], l[r0:r0 + mb],
                               tgt[r0:r0 + mb], v0, ignore_index)
        lse = m + l.log()
        valid = targets != ignore_index
        n_valid = valid.sum().clamp(min=1)
        loss = torch.where(valid, lse - tgt,
                           torch.zeros((), device=dev)).sum() \
            / n_valid.float()
        dx = dw = None
        if need_grads:
            dx = torch.zeros_like(h2)
            if w.requires_grad:
                dw = torch.zeros_like(w)
            for r0 in range(0, M, mb):
                hr = h2[r0:r0 + mb]
                tr = targets[r0:r0 + mb]
                lr = lse[r0:r0 + mb]
                dxr = dx[r0:r0 + mb]
                for v0 in range(0, V, ch):
                    wc = w[v0:v0 + ch]
                    lc = torch.nn.functional.linear(hr, wc)
                    dl = xent_dlogits(lc, tr, lr, v0, ignore_index)
                    dxr.addmm_(dl, wc)
                    if dw is not None:
                        dw[v0:v0 + ch].addmm_(dl.t(), hr)
        ctx.save_for_backward(dx, dw, n_valid)
        return loss

    @staticmethod
    def backward(ctx, gout):
        dx, dw, n_valid = ctx.saved_tensors
        s = (gout.float() / n_valid.float()).to(dx.dtype) \
            if dx is not None else None
        return (dx * s if dx is not None else None, None if dw is None
                else dw * s, None, None, None)


def fused_linear_cross_entropy(h2, w, targets, ignore_index: int = -100):
    need = torch.is_grad_enabled() and (h2.requires_grad or
                                        w.requires_grad)
    return FusedLinearCrossEntropy.apply(h2, w, targets, ignore_index,
                                         need)


class PairedFrozenGemm(torch.autograd.Function):
    """(x@Wg^T, x@Wu^T) for two frozen weights sharing one input (the
    MLP gate/up pair): backward ACCUMULATES the two dgrads into one
    buffer with addmm_ (beta=1 GEMM epilogue), eliminating the autograd
    grad-add that a shared input otherwise costs per layer."""

    @staticmethod
    def forward(ctx, x, wg, wu):
        xs = x.shape
        x2 = x.reshape(-1, xs[-1])
        g = torch.nn.functional.linear(x2, wg)
        u = torch.nn.functional.linear(x2, wu)
        ctx.save_for_backward(wg, wu)
        ctx.xshape = xs
        n = wg.shape[0]
        return (g.reshape(*xs[:-1], n), u.reshape(*xs[:-1], n))

    @staticmethod
    def backward(ctx, dg, du):
        wg, wu = ctx.saved_tensors
        dg2 = dg.reshape(-1, dg.shape[-1])
        du2 = du.reshape(-1, du.shape[-1])
        dx = dg2 @ wg
        dx.addmm_(du2, wu)              # accumulate in the GEMM epilogue
        return dx.reshape(ctx.xshape), None, None


class QKVProj(torch.autograd.Function):
    """q/k/v projections as ONE autograd node (k frozen; q/v LoRA):
    forward runs the three base GEMMs + the fused LoRA low-rank pairs;
    backward accumulates all three dgrads (and the LoRA dx terms, which
    were already in-place) into ONE dx buffer via addmm_ — removing the
    two [M,E] grad-adds autograd inserts per decoder layer for the
    shared attention input."""

    @staticmethod
    def forward(ctx, x, wq, aq, bq, wk, wv, av, bv, scale,
                seed_q, seed_v, keep):
        xs = x.shape
        x2 = x.reshape(-1, xs[-1])
        yq = torch.nn.functional.linear(x2, wq)
        yk = torch.nn.functional.linear(x2, wk)
        yv = torch.nn.functional.linear(x2, wv)
        tq = lora_contract(x2, aq, None, seed_q, keep)
        lora_expand_add(yq, tq, bq, scale)
        tv = lora_contract(x2, av, None, seed_v, keep)
        lora_expand_add(yv, tv, bv, scale)
        ctx.save_for_backward(x2, wq, aq, bq, wk, wv, av, bv, tq, tv)
        ctx.meta = (scale, seed_q, seed_v, keep, xs)
        return (yq.reshape(*xs[:-1], wq.shape[0]),
                yk.reshape(*xs[:-1], wk.shape[0]),
                yv.reshape(*xs[:-1], wv.shape[0]))

    @staticmethod
    def backward(ctx, dyq, dyk, dyv):
        from . import lora_wgrad
        x2, wq, aq, bq, wk, wv, av, bv, tq, tv = ctx.saved_tensors
        scale, seed_q, seed_v, keep, xs = ctx.meta
        dq2 = dyq.reshape(-1, dyq.shape[-1]).contiguous()
        dk2 = dyk.reshape(-1, dyk.shape[-1]).contiguous()
        dv2 = dyv.reshape(-1, dyv.shape[-1]).contiguous()
        dx = dq2 @ wq
        dx.addmm_(dk2, wk)
        dx.addmm_(dv2, wv)
        # LoRA grads + low-rank dx terms (expand_add is in-place on dx)
        dtq = lora_contract(dq2, bq.t().contiguous())
        daq = lora_wgrad(dtq, x2, scale, None, seed_q, keep)
        dbq = lora_wgrad(tq, dq2, scale).t().contiguous()
        lora_expand_add(dx, dtq, aq.t().contiguous(), scale, None,
                        seed_q, keep)
        dtv = lora_contract(dv2, bv.t().contiguous())
        dav = lora_wgrad(dtv, x2, scale, None, seed_v, keep)
        dbv = lora_wgrad(tv, dv2, scale).t().contiguous()
        lora_expand_add(dx, dtv, av.t().contiguous(), scale, None,
                        seed_v, keep)
        return (dx.reshape(xs), None, daq.to(aq.dtype),
                dbq.to(bq.dtype), None, None, dav.to(av.dtype),
                dbv.to(bv.dtype), None, None, None, None)
