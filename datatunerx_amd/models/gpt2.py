"""GPT-2-small-shaped model for the CPU plumbing config (BASELINE.json
configs[0]): proves the CR cascade + trainer loop with zero GPU code.

Faithful GPT-2 block structure (learned positions, pre-LN LayerNorm,
GELU MLP, causal attention) with LoRA on q/v projections so the adapter
checkpoint path is exercised end-to-end on CPU.
"""

from __future__ import annotations

from dataclasses import dataclass

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops.autograd import attention, cross_entropy
from .lora import FrozenLinear, LoRALinearModule


@dataclass
class GPT2Config:
    # padded from GPT-2's 50257 to the next multiple of 8: the
    # fused CE kernel wants V % 8 == 0, ids stay < 50257, and the
    # pad logits train to -inf-ish harmlessly (random-init path;
    # there is no HF-GPT-2 weight loader to stay byte-compatible
    # with)
    vocab_size: int = 50264
    hidden_size: int = 768
    num_hidden_layers: int = 12
    num_attention_heads: int = 12
    max_position_embeddings: int = 1024
    layer_norm_eps: float = 1e-5
    lora_r: int = 8
    lora_alpha: float = 32.0
    lora_dropout: float = 0.0
    lora_targets: tuple = ("q_proj", "v_proj")

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    @classmethod
    def small(cls, **kw):
        return cls(**kw)

    @classmethod
    def tiny(cls, **kw):
        d = dict(vocab_size=512, hidden_size=64, num_hidden_layers=2,
                 num_attention_heads=4, max_position_embeddings=128)
        d.update(kw)
        return cls(**d)


def _proj(cfg, name, in_f, out_f, dtype):
    if name in cfg.lora_targets:
        return LoRALinearModule(in_f, out_f, r=cfg.lora_r,
                                alpha=cfg.lora_alpha,
                                dropout=cfg.lora_dropout, dtype=dtype)
    return FrozenLinear(in_f, out_f, dtype=dtype)


class GPT2Block(nn.Module):
    def __init__(self, cfg: GPT2Config, dtype):
        super().__init__()
        h = cfg.hidden_size
        self.cfg = cfg
        self.ln_1 = nn.LayerNorm(h, eps=cfg.layer_norm_eps, dtype=dtype)
        self.ln_2 = nn.LayerNorm(h, eps=cfg.layer_norm_eps, dtype=dtype)
        for ln in (self.ln_1, self.ln_2):
            ln.weight.requires_grad_(False)
            ln.bias.requires_grad_(False)
        self.q_proj = _proj(cfg, "q_proj", h, h, dtype)
        self.k_proj = _proj(cfg, "k_proj", h, h, dtype)
        self.v_proj = _proj(cfg, "v_proj", h, h, dtype)
        self.o_proj = _proj(cfg, "o_proj", h, h, dtype)
        self.mlp_fc = FrozenLinear(h, 4 * h, dtype=dtype)
        self.mlp_proj = FrozenLinear(4 * h, h, dtype=dtype)

    def forward(self, x):
        B, S, _ = x.shape
        H, D = self.cfg.num_attention_heads, self.cfg.head_dim
        hx = self.ln_1(x)
        q = self.q_proj(hx).view(B, S, H, D)
        k = self.k_proj(hx).view(B, S, H, D)
        v = self.v_proj(hx).view(B, S, H, D)
        o = attention(q, k, v, causal=True)      # BSHD in/out
        o = o.reshape(B, S, H * D)
        x = x + self.o_proj(o)
        hx = self.ln_2(x)
        return x + self.mlp_proj(F.gelu(self.mlp_fc(hx)))


class GPT2ForCausalLM(nn.Module):
    def __init__(self, cfg: GPT2Config, dtype=torch.float32):
        super().__init__()
        self.cfg = cfg
        self.wte = nn.Embedding(cfg.vocab_size, cfg.hidden_size, dtype=dtype)
        self.wpe = nn.Embedding(cfg.max_position_embeddings, cfg.hidden_size,
                                dtype=dtype)
        self.wte.weight.requires_grad_(False)
        self.wpe.weight.requires_grad_(False)
        self.blocks = nn.ModuleList(GPT2Block(cfg, dtype)
                                    for _ in range(cfg.num_hidden_layers))
        self.ln_f = nn.LayerNorm(cfg.hidden_size, eps=cfg.layer_norm_eps,
                                 dtype=dtype)
        self.ln_f.weight.requires_grad_(False)
        self.ln_f.bias.requires_grad_(False)
        self.lm_head = FrozenLinear(cfg.hidden_size, cfg.vocab_size,
                                    dtype=dtype)

    @torch.no_grad()
    def init_random(self, std: float = 0.02, seed: int = 0):
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

    def forward(self, input_ids, labels=None):
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device)
        x = self.wte(input_ids) + self.wpe(pos).unsqueeze(0)
        for blk in self.blocks:
            x = blk(x)
        x = self.ln_f(x)
        if labels is None:
            return self.lm_head(x)
        h = x[:, :-1, :].reshape(-1, self.cfg.hidden_size)
        logits = self.lm_head(h)
        targets = labels[:, 1:].reshape(-1)
        return cross_entropy(logits, targets, ignore_index=-100)

    def sequence_logprobs(self, input_ids, labels):
        """[B] sum log p(target tokens) — DPO building block (same
        contract as LlamaForCausalLM.sequence_logprobs)."""
        from ..ops.autograd import per_row_cross_entropy
        B = input_ids.shape[0]
        logits = self.forward(input_ids)
        h = logits[:, :-1, :].reshape(-1, self.cfg.vocab_size)
        targets = labels[:, 1:].reshape(-1)
        nll = per_row_cross_entropy(h.contiguous(), targets,
                                    ignore_index=-100)
        return -nll.view(B, -1).sum(dim=1)

    def trainable_parameters(self):
        return [(n, p) for n, p in self.named_parameters() if p.requires_grad]
