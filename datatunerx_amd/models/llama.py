"""Native Llama-family decoder built directly on the fused MI355X ops.

Re-implements what the reference obtains from HF `LlamaForCausalLM`
(cmd/tuning/train.py:236-254) as an MI355X-first module stack: plain
bf16 GEMMs through hipBLASLt (torch.matmul) for the frozen projections,
hand-written HIP kernels for RMSNorm, RoPE, flash-attention, SwiGLU and
the fused LoRA path, fused log-softmax cross-entropy with -100 masking
for the loss. No HF dependency in the training hot path.
"""

from __future__ import annotations

import os
from dataclasses import dataclass

import torch
import torch.nn as nn

from ..ops import rope_tables
from ..ops.autograd import (PairedFrozenGemm, QKVProj, attention,
                            cross_entropy, rmsnorm, rmsnorm_tap,
                            rope, swiglu)
from .lora import FrozenLinear, LoRALinearModule


@dataclass
class LlamaConfig:
    vocab_size: int = 32000
    hidden_size: int = 4096
    intermediate_size: int = 11008
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 32
    max_position_embeddings: int = 4096
    rms_norm_eps: float = 1e-5
    rope_theta: float = 10000.0
    # Qwen2-family: bias on q/k/v projections (o/mlp stay bias-free)
    attention_bias: bool = False
    # Qwen3-family: per-head RMSNorm on q and k (weight [head_dim],
    # applied BEFORE RoPE)
    qk_norm: bool = False
    # LoRA (reference defaults: parser.py:138-149, finetune_controller.go:482)
    lora_r: int = 8
    lora_alpha: float = 32.0
    lora_dropout: float = 0.0
    lora_targets: tuple = ("q_proj", "v_proj")
    gradient_checkpointing: bool = False
    # set when heads are tensor-parallel-sharded (head_dim no longer
    # derivable from hidden_size / local heads)
    head_dim_override: int = 0

    @property
    def head_dim(self) -> int:
        if self.head_dim_override:
            return self.head_dim_override
        return self.hidden_size // self.num_attention_heads

    @classmethod
    def llama2_7b(cls, **kw):
        return cls(**kw)

    @classmethod
    def llama2_13b(cls, **kw):
        d = dict(hidden_size=5120, intermediate_size=13824,
                 num_hidden_layers=40, num_attention_heads=40,
                 num_key_value_heads=40)
        d.update(kw)
        return cls(**d)

    @classmethod
    def llama3_8b(cls, **kw):
        """GQA config (32 q heads / 8 kv heads), Llama-3-8B geometry."""
        d = dict(vocab_size=128256, hidden_size=4096,
                 intermediate_size=14336, num_hidden_layers=32,
                 num_attention_heads=32, num_key_value_heads=8,
                 max_position_embeddings=8192, rope_theta=500000.0)
        d.update(kw)
        return cls(**d)

    @classmethod
    def tiny(cls, **kw):
        """CPU-testable config."""
        d = dict(vocab_size=512, hidden_size=128, intermediate_size=256,
                 num_hidden_layers=2, num_attention_heads=4,
                 num_key_value_heads=4, max_position_embeddings=256)
        d.update(kw)
        return cls(**d)

    @classmethod
    def mini(cls, **kw):
        """GPU-capable small config (head_dim 64 — the attention
        kernels support D in {64, 128}); for on-hardware control-plane
        and smoke tests."""
        d = dict(vocab_size=512, hidden_size=256, intermediate_size=512,
                 num_hidden_layers=2, num_attention_heads=4,
                 num_key_value_heads=4, max_position_embeddings=512)
        d.update(kw)
        return cls(**d)


def _proj(cfg: LlamaConfig, name: str, in_f: int, out_f: int, lora: bool,
          dtype):
    if lora and name in cfg.lora_targets:
        return LoRALinearModule(in_f, out_f, r=cfg.lora_r,
                                alpha=cfg.lora_alpha,
                                dropout=cfg.lora_dropout, dtype=dtype)
    return FrozenLinear(in_f, out_f, dtype=dtype)


class LlamaAttention(nn.Module):
    def __init__(self, cfg: LlamaConfig, lora: bool, dtype):
        super().__init__()
        self.cfg = cfg
        H, D = cfg.num_attention_heads, cfg.head_dim
        Hkv = cfg.num_key_value_heads
        self.q_proj = _proj(cfg, "q_proj", cfg.hidden_size, H * D, lora, dtype)
        self.k_proj = _proj(cfg, "k_proj", cfg.hidden_size, Hkv * D, lora, dtype)
        self.v_proj = _proj(cfg, "v_proj", cfg.hidden_size, Hkv * D, lora, dtype)
        self.o_proj = _proj(cfg, "o_proj", H * D, cfg.hidden_size, lora, dtype)
        if cfg.attention_bias:
            # Qwen2 convention: bias on q/k/v only. Added as a separate
            # broadcast add after the (possibly fused) projections —
            # frozen under LoRA, trainable under train_base like every
            # other base weight.
            self.q_bias = nn.Parameter(torch.zeros(H * D, dtype=dtype),
                                       requires_grad=False)
            self.k_bias = nn.Parameter(torch.zeros(Hkv * D, dtype=dtype),
                                       requires_grad=False)
            self.v_bias = nn.Parameter(torch.zeros(Hkv * D, dtype=dtype),
                                       requires_grad=False)
        else:
            self.q_bias = self.k_bias = self.v_bias = None
        if cfg.qk_norm:
            # Qwen3 qk-norm: same fused rmsnorm kernel, rows = every
            # head vector ([B*S*H, D] through the last-dim-generic
            # launcher)
            self.q_norm = nn.Parameter(torch.ones(D, dtype=dtype),
                                       requires_grad=False)
            self.k_norm = nn.Parameter(torch.ones(D, dtype=dtype),
                                       requires_grad=False)
        else:
            self.q_norm = self.k_norm = None

    def _fused_qkv(self) -> bool:
        """One autograd node for q/k/v (backward dgrads accumulate via
        addmm_ instead of autograd's per-consumer grad adds): the
        standard LoRA shape — q/v LoRA (same scale, fused-RNG-capable),
        k frozen, base weights untrained."""
        qm, km, vm = self.q_proj, self.k_proj, self.v_proj
        return (type(km) is FrozenLinear and
                isinstance(qm, LoRALinearModule) and
                isinstance(vm, LoRALinearModule) and
                not km.weight.requires_grad and
                not qm.weight.requires_grad and
                qm.scale == vm.scale and qm.r == vm.r and
                qm.r <= 16 and qm.in_features % 8 == 0 and
                qm.dropout == vm.dropout)

    def forward(self, x, cos, sin, pos0: int = 0, kv_cache=None,
                pos_dev=None, residual=None):
        B, S, _ = x.shape
        cfg = self.cfg
        H, Hkv, D = cfg.num_attention_heads, cfg.num_key_value_heads, cfg.head_dim
        if self._fused_qkv():
            qm, km, vm = self.q_proj, self.k_proj, self.v_proj
            keep, sq, sv = 1.0, 0, 0
            if self.training and qm.dropout > 0.0:
                keep = 1.0 - qm.dropout
                sq = int(torch.randint(0, 2 ** 31 - 1, (1,)).item())
                sv = int(torch.randint(0, 2 ** 31 - 1, (1,)).item())
            q, k, v = QKVProj.apply(x, qm.weight, qm.lora_A, qm.lora_B,
                                    km.weight, vm.weight, vm.lora_A,
                                    vm.lora_B, qm.scale, sq, sv, keep)
        else:
            q = self.q_proj(x)
            k = self.k_proj(x)
            v = self.v_proj(x)
        if self.q_bias is not None:
            q = q + self.q_bias
            k = k + self.k_bias
            v = v + self.v_bias
        q = q.view(B, S, H, D)
        k = k.view(B, S, Hkv, D)
        v = v.view(B, S, Hkv, D)
        if self.q_norm is not None:
            q = rmsnorm(q.contiguous(), self.q_norm, cfg.rms_norm_eps)
            k = rmsnorm(k.contiguous(), self.k_norm, cfg.rms_norm_eps)
        q = rope(q, cos, sin, pos0, pos_dev)
        k = rope(k, cos, sin, pos0, pos_dev)
        len_dev = None
        if kv_cache is not None:
            k, v, len_dev = kv_cache.update(k, v)   # serving path (BSHD)
        o = attention(q, k, v, causal=True, len_dev=len_dev)
        o = o.reshape(B, S, H * D)
        if residual is not None and isinstance(self.o_proj, FrozenLinear):
            return self.o_proj(o, residual)   # residual in GEMM epilogue
        y = self.o_proj(o)
        return y if residual is None else residual + y


class LlamaMLP(nn.Module):
    def __init__(self, cfg: LlamaConfig, lora: bool, dtype):
        super().__init__()
        self.gate_proj = _proj(cfg, "gate_proj", cfg.hidden_size,
                               cfg.intermediate_size, lora, dtype)
        self.up_proj = _proj(cfg, "up_proj", cfg.hidden_size,
                             cfg.intermediate_size, lora, dtype)
        self.down_proj = _proj(cfg, "down_proj", cfg.intermediate_size,
                               cfg.hidden_size, lora, dtype)

    def forward(self, x, residual=None):
        if type(self.gate_proj) is FrozenLinear and \
                type(self.up_proj) is FrozenLinear and \
                not self.gate_proj.weight.requires_grad:
            # one node for the gate/up pair: backward accumulates both
            # dgrads with addmm_ instead of a grad-add per layer
            g, u = PairedFrozenGemm.apply(x, self.gate_proj.weight,
                                          self.up_proj.weight)
            h = swiglu(g, u)
        else:
            h = swiglu(self.gate_proj(x), self.up_proj(x))
        if residual is not None and isinstance(self.down_proj, FrozenLinear):
            return self.down_proj(h, residual)    # residual in epilogue
        y = self.down_proj(h)
        return y if residual is None else residual + y


class LlamaDecoderLayer(nn.Module):
    def __init__(self, cfg: LlamaConfig, lora: bool, dtype):
        super().__init__()
        self.cfg = cfg
        self.input_layernorm = nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=dtype), requires_grad=False)
        self.post_attention_layernorm = nn.Parameter(
            torch.ones(cfg.hidden_size, dtype=dtype), requires_grad=False)
        self.self_attn = LlamaAttention(cfg, lora, dtype)
        self.mlp = LlamaMLP(cfg, lora, dtype)

    def forward(self, x, cos, sin, pos0: int = 0, kv_cache=None,
                pos_dev=None):
        h, res = rmsnorm_tap(x, self.input_layernorm,
                             self.cfg.rms_norm_eps)
        x = self.self_attn(h, cos, sin, pos0, kv_cache, pos_dev,
                           residual=res)
        h, res = rmsnorm_tap(x, self.post_attention_layernorm,
                             self.cfg.rms_norm_eps)
        return self.mlp(h, residual=res)


class LlamaForCausalLM(nn.Module):
    """`lora=True` trains only adapters (reference's finetuning_type
    lora); `train_base=True` makes every weight trainable (full-param
    SFT, the reference's finetuning_type=full — parser.py:131-137)."""

    def __init__(self, cfg: LlamaConfig, lora: bool = True,
                 dtype=torch.bfloat16, train_base: bool = False):
        super().__init__()
        self.cfg = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size,
                                         dtype=dtype)
        self.embed_tokens.weight.requires_grad_(False)
        self.layers = nn.ModuleList(
            LlamaDecoderLayer(cfg, lora, dtype)
            for _ in range(cfg.num_hidden_layers))
        self.norm = nn.Parameter(torch.ones(cfg.hidden_size, dtype=dtype),
                                 requires_grad=False)
        self.lm_head = FrozenLinear(cfg.hidden_size, cfg.vocab_size,
                                    dtype=dtype)
        cos, sin = rope_tables(cfg.max_position_embeddings, cfg.head_dim,
                               cfg.rope_theta, dtype=torch.float32)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)
        if train_base:
            for p in self.parameters():
                p.requires_grad_(True)

    @torch.no_grad()
    def init_random(self, std: float = 0.02, seed: int = 0):
        """Random-init weights (bench/tests run without checkpoints —
        there is no network for real weights). Generates on the params'
        device (fast on GPU; deterministic for a fixed seed+device)."""
        dev = next(self.parameters()).device
        g = torch.Generator(device=dev).manual_seed(seed)
        for p in self.parameters():
            if p.dim() >= 2:
                p.copy_(torch.randn(p.shape, generator=g, device=dev,
                                    dtype=torch.float32).mul_(std).to(p.dtype))
        for n, p in self.named_parameters():
            if "lora_B" in n:
                p.zero_()
        return self

    def hidden_states(self, input_ids, pos0: int = 0, kv_caches=None,
                      pos_dev=None):
        x = self.embed_tokens(input_ids)
        cos, sin = self.rope_cos, self.rope_sin
        for i, layer in enumerate(self.layers):
            cache = kv_caches[i] if kv_caches is not None else None
            if self.cfg.gradient_checkpointing and self.training:
                x = torch.utils.checkpoint.checkpoint(
                    layer, x, cos, sin, pos0, cache, use_reentrant=False)
            else:
                x = layer(x, cos, sin, pos0, cache, pos_dev)
        return rmsnorm(x, self.norm, self.cfg.rms_norm_eps)

    def forward(self, input_ids, labels=None, pos0: int = 0,
                kv_caches=None, pos_dev=None):
        """Returns loss (if labels given, shifted CE with -100 ignore)
        else logits."""
        h = self.hidden_states(input_ids, pos0, kv_caches, pos_dev)
        if labels is None:
            return self.lm_head(h)
        if os.environ.get("DTX_FUSED_CE") == "1" and \
                self.cfg.vocab_size % 8 == 0:
            # chunked-vocab fused lm_head+CE: no [M,V] logits tensor
            # (frees 1.6 GB at V=32k / 6.3 GB at V=128k, mb24 seq1024).
            # OPT-IN: measured 0.5-2% slower per step than the
            # materialized path (the chunk GEMMs + per-chunk dx
            # accumulation cost more than the logits HBM traffic they
            # save), so the default optimizes time; set DTX_FUSED_CE=1
            # when activation memory is the constraint.
            # Keep ALL positions (h stays the contiguous buffer — no
            # [M,E] copy) and mark each row's LAST position ignored
            # instead of slicing the shift.
            from ..ops.autograd import fused_linear_cross_entropy
            t = torch.full_like(labels, -100)
            t[:, :-1] = labels[:, 1:]
            return fused_linear_cross_entropy(
                h.reshape(-1, self.cfg.hidden_size),
                self.lm_head.weight, t.reshape(-1), ignore_index=-100)
        # shift: predict token t+1 from position t
        h = h[:, :-1, :].reshape(-1, self.cfg.hidden_size)
        targets = labels[:, 1:].reshape(-1)
        logits = self.lm_head(h)
        return cross_entropy(logits, targets, ignore_index=-100)

    def sequence_logprobs(self, input_ids, labels):
        """[B] sum of log p(target token) over each row's non-ignored
        (shifted) labels — the DPO building block. Gradients flow to
        trainable params; runs the same masked xent kernels as the
        loss path, just unreduced."""
        from ..ops.autograd import per_row_cross_entropy
        B = input_ids.shape[0]
        h = self.hidden_states(input_ids)
        h = h[:, :-1, :].reshape(-1, self.cfg.hidden_size)
        targets = labels[:, 1:].reshape(-1)
        logits = self.lm_head(h)
        nll = per_row_cross_entropy(logits, targets, ignore_index=-100)
        return -nll.view(B, -1).sum(dim=1)

    def trainable_parameters(self):
        return [(n, p) for n, p in self.named_parameters() if p.requires_grad]
