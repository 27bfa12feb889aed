"""Plain-PyTorch reference implementations of every hot op.

These are the numerical contract for the hand-written gfx950 HIP kernels
in datatunerx_amd/ops/hip/: each GPU test compares the HIP kernel against
the fp32 version of the op here. They also serve as the CPU compute path
(tests, the GPT-2-small plumbing config) — on a GPU box the HIP extension
is mandatory and these are NOT silently used (see ops/__init__.py).

Workloads mirror what the reference exercises through HF/PEFT/DeepSpeed
(reference: SURVEY.md §2.4; cmd/tuning/train.py:236-280).
"""

from __future__ import annotations

import torch


# ---------------------------------------------------------------- RMSNorm
def rmsnorm_fwd(x: torch.Tensor, w: torch.Tensor, eps: float = 1e-5):
    """y = x * rsqrt(mean(x^2) + eps) * w.  Returns (y, inv_rms[f32])."""
    xf = x.float()
    inv = torch.rsqrt(xf.pow(2).mean(dim=-1) + eps)  # [M]
    y = (xf * inv.unsqueeze(-1)) * w.float()
    return y.to(x.dtype), inv


def rmsnorm_bwd(dy: torch.Tensor, x: torch.Tensor, w: torch.Tensor,
                inv: torch.Tensor):
    """Returns (dx, dw[f32]). dw is reduced over all rows."""
    xf, dyf, wf = x.float(), dy.float(), w.float()
    H = xf.shape[-1]
    g = dyf * wf                                     # [M,H]
    # dx = inv*g - inv^3/H * (g·x) * x
    dot = (g * xf).sum(dim=-1, keepdim=True)         # [M,1]
    dx = inv.unsqueeze(-1) * g - (inv.unsqueeze(-1) ** 3 / H) * dot * xf
    dw = (dyf * xf * inv.unsqueeze(-1)).sum(dim=0)
    return dx.to(x.dtype), dw


# ------------------------------------------------------------------- RoPE
def rope_tables(seq_len: int, head_dim: int, base: float = 10000.0,
                device=None, dtype=torch.float32):
    """cos/sin tables [S, D/2] (host-precomputed; GPU kernels load these)."""
    inv_freq = 1.0 / (base ** (torch.arange(0, head_dim, 2,
                                            device=device).float() / head_dim))
    t = torch.arange(seq_len, device=device).float()
    freqs = torch.outer(t, inv_freq)                 # [S, D/2]
    return freqs.cos().to(dtype), freqs.sin().to(dtype)


def rope_fwd(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
             pos0: int = 0):
    """Llama rotate-half RoPE. x: [B, S, H, D]; cos/sin: [>=pos0+S, D/2]."""
    B, S, H, D = x.shape
    xf = x.float()
    x1, x2 = xf[..., : D // 2], xf[..., D // 2:]
    c = cos[pos0: pos0 + S].float().view(1, S, 1, D // 2)
    s = sin[pos0: pos0 + S].float().view(1, S, 1, D // 2)
    y1 = x1 * c - x2 * s
    y2 = x2 * c + x1 * s
    return torch.cat([y1, y2], dim=-1).to(x.dtype)


def rope_bwd(dy: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
             pos0: int = 0):
    """Gradient of rope_fwd = rotation by -theta (transpose of rotation)."""
    B, S, H, D = dy.shape
    df = dy.float()
    d1, d2 = df[..., : D // 2], df[..., D // 2:]
    c = cos[pos0: pos0 + S].float().view(1, S, 1, D // 2)
    s = sin[pos0: pos0 + S].float().view(1, S, 1, D // 2)
    x1 = d1 * c + d2 * s
    x2 = -d1 * s + d2 * c
    return torch.cat([x1, x2], dim=-1).to(dy.dtype)


# ----------------------------------------------------------------- SwiGLU
def swiglu_fwd(gate: torch.Tensor, up: torch.Tensor):
    g = gate.float()
    return (g * torch.sigmoid(g) * up.float()).to(gate.dtype)


def swiglu_bwd(dout: torch.Tensor, gate: torch.Tensor, up: torch.Tensor):
    g, u, d = gate.float(), up.float(), dout.float()
    sig = torch.sigmoid(g)
    silu = g * sig
    dgate = d * u * (sig + silu * (1.0 - sig))
    dup = d * silu
    return dgate.to(gate.dtype), dup.to(up.dtype)


# ----------------------------------------------- fused softmax cross-entropy
def softmax_xent_fwd(logits: torch.Tensor, targets: torch.Tensor,
                     ignore_index: int = -100):
    """Per-row CE with ignore mask. Returns (loss[f32, N], lse[f32, N]).

    loss[i] = lse[i] - logit[i, t_i]  (0 where ignored).
    Matches the reference's -100 masking (cmd/tuning/train.py:50,98-103).
    """
    lf = logits.float()
    lse = torch.logsumexp(lf, dim=-1)
    mask = targets != ignore_index
    safe_t = targets.masked_fill(~mask, 0)
    picked = lf.gather(-1, safe_t.unsqueeze(-1)).squeeze(-1)
    loss = torch.where(mask, lse - picked, torch.zeros_like(lse))
    return loss, lse


def softmax_xent_bwd(logits: torch.Tensor, targets: torch.Tensor,
                     lse: torch.Tensor, dloss: torch.Tensor,
                     ignore_index: int = -100):
    """dlogits = dloss[i] * (softmax - onehot), 0 on ignored rows."""
    lf = logits.float()
    p = torch.exp(lf - lse.unsqueeze(-1))
    mask = (targets != ignore_index)
    safe_t = targets.masked_fill(~mask, 0)
    p.scatter_add_(-1, safe_t.unsqueeze(-1),
                   -torch.ones_like(p[..., :1]))
    p = p * (dloss * mask.float()).unsqueeze(-1)
    return p.to(logits.dtype)


# -------------------------------------------------------- flash attention
def attn_fwd(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
             causal: bool = True, scale: float | None = None):
    """BSHD public contract: q [B,S,Hq,D], k/v [B,Skv,Hkv,D].

    Returns (o [B,S,Hq,D], lse [f32, B,Hq,S])."""
    o, lse = _attn_fwd_bhsd(q.permute(0, 2, 1, 3), k.permute(0, 2, 1, 3),
                            v.permute(0, 2, 1, 3), causal, scale)
    return o.permute(0, 2, 1, 3).contiguous(), lse


def attn_bwd(q, k, v, o, do, lse, causal: bool = True,
             scale: float | None = None):
    """BSHD public contract; returns (dq, dk, dv) in BSHD."""
    p = lambda t: t.permute(0, 2, 1, 3)
    dq, dk, dv = _attn_bwd_bhsd(p(q), p(k), p(v), p(o), p(do), lse,
                                causal, scale)
    return (p(dq).contiguous(), p(dk).contiguous(), p(dv).contiguous())


def _attn_fwd_bhsd(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                   causal: bool = True, scale: float | None = None):
    """q: [B,Hq,S,D], k/v: [B,Hkv,S,D] (GQA by head repeat).

    Returns (o [B,Hq,S,D], lse [f32, B,Hq,S]). fp32 math.
    """
    B, Hq, S, D = q.shape
    Hkv = k.shape[1]
    if scale is None:
        scale = 1.0 / (D ** 0.5)
    if Hkv != Hq:
        rep = Hq // Hkv
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    s = torch.einsum("bhqd,bhkd->bhqk", q.float(), k.float()) * scale
    if causal:
        Skv = k.shape[2]
        mask = torch.ones(S, Skv, dtype=torch.bool, device=q.device).tril(
            diagonal=Skv - S)
        s = s.masked_fill(~mask, float("-inf"))
    lse = torch.logsumexp(s, dim=-1)
    p = torch.exp(s - lse.unsqueeze(-1))
    o = torch.einsum("bhqk,bhkd->bhqd", p, v.float())
    return o.to(q.dtype), lse


def _attn_bwd_bhsd(q, k, v, o, do, lse, causal: bool = True,
                   scale: float | None = None):
    """Returns (dq, dk, dv) with GQA reduction over repeated heads."""
    B, Hq, S, D = q.shape
    Hkv = k.shape[1]
    if scale is None:
        scale = 1.0 / (D ** 0.5)
    rep = Hq // Hkv
    kk = k.repeat_interleave(rep, dim=1) if Hkv != Hq else k
    vv = v.repeat_interleave(rep, dim=1) if Hkv != Hq else v
    qf, kf, vf, of, dof = (t.float() for t in (q, kk, vv, o, do))
    s = torch.einsum("bhqd,bhkd->bhqk", qf, kf) * scale
    if causal:
        Skv = kf.shape[2]
        mask = torch.ones(S, Skv, dtype=torch.bool, device=q.device).tril(
            diagonal=Skv - S)
        s = s.masked_fill(~mask, float("-inf"))
    p = torch.exp(s - lse.unsqueeze(-1))
    dv = torch.einsum("bhqk,bhqd->bhkd", p, dof)
    dp = torch.einsum("bhqd,bhkd->bhqk", dof, vf)
    delta = (dof * of).sum(dim=-1, keepdim=True)     # [B,H,S,1]
    ds = p * (dp - delta) * scale
    dq = torch.einsum("bhqk,bhkd->bhqd", ds, kf)
    dk = torch.einsum("bhqk,bhqd->bhkd", ds, qf)
    if Hkv != Hq:
        dk = dk.view(B, Hkv, rep, S, D).sum(dim=2)
        dv = dv.view(B, Hkv, rep, S, D).sum(dim=2)
    return dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype)


# ------------------------------------------------------------- LoRA pieces
# Optional `mask` mirrors the fused-dropout GPU kernels: the masked
# element enters the product in the INPUT dtype (x*mask rounded once).
def _masked(x, mask):
    if mask is None:
        return x
    return (x * mask).to(x.dtype)


def dropout_mask(M: int, K: int, seed: int, keep: float,
                 device=None, dtype=torch.bfloat16):
    """Bit-exact CPU twin of the GPU counter-based dropout RNG
    (splitmix64 over 4-element groups, 16 random bits/element,
    threshold keep*65536, value = bf16-rounded 1/keep). Lets CPU
    fallbacks and tests reproduce the fused kernels' mask exactly."""
    assert (M * K) % 4 == 0
    n_groups = (M * K) // 4

    def _u(c):                       # unsigned 64-bit const -> torch int64
        return torch.tensor(c - (1 << 64) if c >= (1 << 63) else c,
                            dtype=torch.int64, device=device)

    def _lshr(z, k):                 # logical shift right on int64
        return (z >> k) & ((1 << (64 - k)) - 1)

    g = torch.arange(n_groups, dtype=torch.int64, device=device)
    z = _u(seed & ((1 << 64) - 1)) + g * _u(0x9E3779B97F4A7C15)
    z = z ^ _lshr(z, 30)
    z = z * _u(0xBF58476D1CE4E5B9)
    z = z ^ _lshr(z, 27)
    z = z * _u(0x94D049BB133111EB)
    z = z ^ _lshr(z, 31)
    thr16 = min(65536, int(keep * 65536.0 + 0.5))
    bits = torch.stack([(z >> (16 * i)) & 0xFFFF for i in range(4)], dim=1)
    keepers = bits < thr16
    inv_keep = torch.tensor(1.0 / keep).to(torch.bfloat16).item()
    return (keepers.reshape(M, K).to(dtype) * inv_keep).to(dtype)


def _resolve_mask(mask, seed, keep, shape, device, dtype):
    if mask is not None or keep >= 1.0:
        return mask
    return dropout_mask(shape[0], shape[1], seed, keep, device=device,
                        dtype=dtype)


def lora_contract(x: torch.Tensor, w: torch.Tensor, mask=None,
                  seed: int = 0, keep: float = 1.0):
    """t[M,r] = (x o mask)[M,K] @ w[r,K]^T  (fp32 out). w is HF lora_A
    layout [r,K], or for the dgrad pass w = lora_B transposed."""
    x2 = x.reshape(-1, x.shape[-1])
    mask = _resolve_mask(mask, seed, keep, x2.shape, x.device, x.dtype)
    return _masked(x2, mask).float() @ w.float().t()


def lora_expand_add(y: torch.Tensor, t: torch.Tensor, w: torch.Tensor,
                    scale: float, mask=None, seed: int = 0,
                    keep: float = 1.0):
    """y[M,N] += (mask o) scale * t[M,r] @ w[N,r]^T  (in-place on y).
    HF lora_B layout is [N,r]."""
    y2 = y.reshape(-1, y.shape[-1])
    mask = _resolve_mask(mask, seed, keep, y2.shape, y.device, y.dtype)
    d = scale * (t.float() @ w.float().t())
    if mask is not None:
        d = d * mask.float()
    y2.add_(d.to(y.dtype))
    return y


def lora_wgrad(t: torch.Tensor, x: torch.Tensor, scale: float = 1.0,
               mask=None, seed: int = 0, keep: float = 1.0):
    """dW[r,K] = scale * t[M,r]^T @ (x o mask)[M,K]  (fp32)."""
    x2 = x.reshape(-1, x.shape[-1])
    mask = _resolve_mask(mask, seed, keep, x2.shape, x.device, x.dtype)
    return scale * (t.float().t() @ _masked(x2, mask).float())


# ------------------------------------------------------------ fused AdamW
def adamw_step(p_bf16: torch.Tensor, master: torch.Tensor,
               grad: torch.Tensor, m: torch.Tensor, v: torch.Tensor,
               lr: float, beta1: float, beta2: float, eps: float,
               weight_decay: float, step: int):
    """Single-pass AdamW on flat fp32 master weights; p_bf16 gets the
    rounded copy. Decoupled weight decay (torch AdamW convention)."""
    gf = grad.float()
    m.mul_(beta1).add_(gf, alpha=1.0 - beta1)
    v.mul_(beta2).addcmul_(gf, gf, value=1.0 - beta2)
    bc1 = 1.0 - beta1 ** step
    bc2 = 1.0 - beta2 ** step
    master.mul_(1.0 - lr * weight_decay)
    denom = (v / bc2).sqrt_().add_(eps)
    master.addcdiv_(m, denom, value=-lr / bc1)
    p_bf16.copy_(master.to(p_bf16.dtype))


def l2_norm(flat: torch.Tensor) -> torch.Tensor:
    return flat.float().pow(2).sum().sqrt()


def xent_lse_merge(logits_c, targets, m_run, l_run, tgt, v0: int,
                   ignore_index: int = -100):
    """CPU fp32 reference of the chunked online-LSE merge (in-place)."""
    x = logits_c.float()
    cmax = x.max(dim=-1).values
    csum = (x - cmax[:, None]).exp().sum(dim=-1)
    m_new = torch.maximum(m_run, cmax)
    l_run.mul_((m_run - m_new).exp()).add_(csum * (cmax - m_new).exp())
    m_run.copy_(m_new)
    vc = x.shape[-1]
    inside = (targets != ignore_index) & (targets >= v0) & \
        (targets < v0 + vc)
    idx = inside.nonzero(as_tuple=True)[0]
    if idx.numel():
        tgt[idx] = x[idx, (targets[idx] - v0)]


def xent_dlogits(logits_c, targets, lse, v0: int,
                 ignore_index: int = -100):
    x = logits_c.float()
    dl = (x - lse[:, None]).exp()
    vc = x.shape[-1]
    inside = (targets != ignore_index) & (targets >= v0) & \
        (targets < v0 + vc)
    idx = inside.nonzero(as_tuple=True)[0]
    if idx.numel():
        dl[idx, (targets[idx] - v0)] -= 1.0
    dl[targets == ignore_index] = 0.0
    return dl.to(logits_c.dtype)
