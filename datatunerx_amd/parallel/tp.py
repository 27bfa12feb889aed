"""Tensor parallelism for the inference-compare service (SURVEY.md §7
step 7: TP-over-RCCL path for the 13B eval service; reference serves on
a single GPU — generate.go:160-209 — so TP is a capability extension).

Sharding (Megatron-style, sized for xGMI):
  - column-parallel: q/k/v projections (by head), gate/up (by column):
    each rank holds out_features / ws rows; no comm on the way in.
  - row-parallel: o_proj / down_proj (by input column): partial outputs
    summed with ONE all-reduce per block over RCCL.
  - lm_head: column-parallel over vocab; logits all-gathered.
  - embeddings and norms replicated.

LoRA adapters shard with their base: B rows for column-parallel,
A columns for row-parallel (exact: (x A^T) B^T splits over B rows;
row-parallel splits the x columns feeding A).

Weights are generated rank-identically (full tensor from the seeded
generator, then sliced) so a TP model is numerically the same model as
the single-GPU one — verified by tests/test_tp_gloo.py on gloo.
"""

from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn

from ..models.llama import LlamaConfig, LlamaForCausalLM
from ..models.lora import FrozenLinear, LoRALinearModule


def _allreduce(x):
    if dist.is_initialized() and dist.get_world_size() > 1:
        dist.all_reduce(x, op=dist.ReduceOp.SUM)
    return x


class _RowParallelOut(torch.autograd.Function):
    """All-reduce in forward (row-parallel output); identity backward
    (serving path only needs forward)."""

    @staticmethod
    def forward(ctx, x):
        return _allreduce(x.contiguous())

    @staticmethod
    def backward(ctx, g):
        return g


def build_tp_llama(cfg: LlamaConfig, tp_rank: int, tp_ws: int,
                   lora: bool = False, dtype=torch.bfloat16,
                   device=None, seed: int = 0):
    """Build rank `tp_rank`'s shard of a Llama TP group.

    Returns a LlamaForCausalLM whose projections are sharded; its
    forward() needs the returned model to run inside an initialized
    process group of size tp_ws (world ranks = tp ranks).
    """
    assert cfg.num_attention_heads % tp_ws == 0, "heads % tp"
    assert cfg.num_key_value_heads % tp_ws == 0, "kv heads % tp"
    assert cfg.intermediate_size % tp_ws == 0, "ffn % tp"
    assert cfg.vocab_size % tp_ws == 0, "vocab % tp"
    device = device or torch.device("cpu")

    # rank-local config: fewer heads / smaller ffn, same head_dim
    import dataclasses
    local = dataclasses.replace(
        cfg,
        num_attention_heads=cfg.num_attention_heads // tp_ws,
        num_key_value_heads=cfg.num_key_value_heads // tp_ws,
        intermediate_size=cfg.intermediate_size // tp_ws,
        head_dim_override=cfg.head_dim)
    with torch.device(device):
        model = LlamaForCausalLM(local, lora=lora, dtype=dtype)
        # vocab-parallel lm_head shard (embed stays replicated)
        model.lm_head = FrozenLinear(cfg.hidden_size,
                                     cfg.vocab_size // tp_ws, dtype=dtype)
    model.cfg_full = cfg
    model.tp_rank, model.tp_ws = tp_rank, tp_ws

    # generate FULL weights deterministically, slice this rank's shard
    _init_tp_weights(model, cfg, tp_rank, tp_ws, seed, device)

    # row-parallel outputs need an all-reduce; wrap o_proj/down_proj
    for layer in model.layers:
        layer.self_attn.o_proj = _RowParallelLinear(layer.self_attn.o_proj)
        layer.mlp.down_proj = _RowParallelLinear(layer.mlp.down_proj)
    model.lm_head = _VocabParallelHead(model.lm_head, tp_ws)
    return model


class _RowParallelLinear(nn.Module):
    def __init__(self, inner):
        super().__init__()
        self.inner = inner

    @property
    def weight(self):
        return self.inner.weight

    def forward(self, x):
        return _RowParallelOut.apply(self.inner(x))


class _VocabParallelHead(nn.Module):
    """Column-parallel lm_head: local logits [*, V/ws] -> all-gather."""

    def __init__(self, inner, tp_ws):
        super().__init__()
        self.inner = inner
        self.tp_ws = tp_ws

    @property
    def weight(self):
        return self.inner.weight

    def forward(self, x):
        local = self.inner(x)
        if self.tp_ws == 1 or not dist.is_initialized():
            return local
        parts = [torch.empty_like(local) for _ in range(self.tp_ws)]
        dist.all_gather(parts, local.contiguous())
        return torch.cat(parts, dim=-1)


@torch.no_grad()
def _init_tp_weights(model, cfg: LlamaConfig, rank: int, ws: int,
                     seed: int, device, std: float = 0.02):
    """Identical numbers to LlamaForCausalLM.init_random(seed) on a
    single GPU: draw each FULL parameter from the same generator order
    (named_parameters of the full model) and keep this rank's slice.

    Shard axes: q/k/v/gate/up weight -> rows; o/down weight -> cols;
    lm_head -> rows; embed/norm -> replicated. LoRA: B rows with
    column-parallel, A cols with row-parallel."""
    g = torch.Generator(device=device).manual_seed(seed)

    # the full model's parameter order/shapes, without materializing it:
    # reconstruct names from the local model + full config.
    H, Hkv = cfg.num_attention_heads, cfg.num_key_value_heads
    D, E, I = cfg.head_dim, cfg.hidden_size, cfg.intermediate_size
    col = {"q_proj": H * D, "k_proj": Hkv * D, "v_proj": Hkv * D,
           "gate_proj": I, "up_proj": I}
    row = {"o_proj": (E, H * D), "down_proj": (E, I)}

    def draw(shape):
        return torch.randn(shape, generator=g, device=device,
                           dtype=torch.float32).mul_(std)

    for name, p in model.named_parameters():
        if p.dim() < 2:
            continue                      # norms stay ones (init_random
            #                               only touches dim >= 2)
        # resolve the module kind from the name
        kind = None
        for k in list(col) + list(row) + ["lm_head", "embed_tokens"]:
            if f"{k}." in name or name.startswith(k):
                kind = k
                break
        is_lora_a = "lora_A" in name
        is_lora_b = "lora_B" in name
        # NOTE: every dim>=2 param of the FULL model must be drawn (even
        # the ones zeroed afterwards) to keep this generator stream
        # bit-identical to single-GPU init_random(seed).
        if kind in col:
            n_full = col[kind]
            sh = n_full // ws
            if is_lora_a:                 # [r, E] replicated
                p.copy_(draw((p.shape[0], E)).to(p.dtype))
            elif is_lora_b:               # [n_full, r] -> rows; zeroed
                draw((n_full, p.shape[1]))
                p.zero_()
            else:                         # [n_full, E] -> rows
                w = draw((n_full, E))
                p.copy_(w[rank * sh:(rank + 1) * sh].to(p.dtype))
        elif kind in row:
            out_f, in_full = row[kind]
            sh = in_full // ws
            if is_lora_a:                 # [r, in_full] -> cols
                w = draw((p.shape[0], in_full))
                p.copy_(w[:, rank * sh:(rank + 1) * sh].to(p.dtype))
            elif is_lora_b:               # [out_f, r] replicated; zeroed
                draw((out_f, p.shape[1]))
                p.zero_()
            else:                         # [out_f, in_full] -> cols
                w = draw((out_f, in_full))
                p.copy_(w[:, rank * sh:(rank + 1) * sh].to(p.dtype))
        elif kind == "lm_head":
            sh = cfg.vocab_size // ws
            w = draw((cfg.vocab_size, E))
            p.copy_(w[rank * sh:(rank + 1) * sh].to(p.dtype))
        elif kind == "embed_tokens":
            p.copy_(draw(tuple(p.shape)).to(p.dtype))
        else:                             # unexpected 2D param: replicate
            p.copy_(draw(tuple(p.shape)).to(p.dtype))
    return model


def load_adapter_tp(model, adapter_dir: str, cfg: LlamaConfig, rank: int,
                    ws: int, prefix: str = "base_model.model."):
    """Load a full HF-PEFT adapter checkpoint into a TP shard."""
    import os

    from safetensors.torch import load_file
    sd = load_file(os.path.join(adapter_dir, "adapter_model.safetensors"))
    sd = shard_adapter_state(sd, cfg, rank, ws)
    # _RowParallelLinear wraps o_proj/down_proj: strip the ".inner" hop
    # so checkpoint module paths still resolve
    mods = {n.replace(".inner", ""): m for n, m in model.named_modules()
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
                # strip the _RowParallelLinear wrapper path if present
                if mod_name in mods:
                    getattr(mods[mod_name], attr).data.copy_(
                        tensor.to(getattr(mods[mod_name], attr).dtype))
                    n_loaded += 1
    return n_loaded


def shard_adapter_state(sd: dict, cfg: LlamaConfig, rank: int, ws: int):
    """Slice a full HF-PEFT adapter state dict for TP rank `rank`.

    Column-parallel targets (q/k/v/gate/up): lora_B rows sharded,
    lora_A replicated. Row-parallel targets (o/down): lora_A columns
    sharded, lora_B replicated."""
    col = ("q_proj", "k_proj", "v_proj", "gate_proj", "up_proj")
    rowp = ("o_proj", "down_proj")
    out = {}
    for key, t in sd.items():
        kind = None
        for k in col + rowp:
            if f"{k}." in key:
                kind = k
                break
        if kind in col and "lora_B" in key:
            sh = t.shape[0] // ws
            out[key] = t[rank * sh:(rank + 1) * sh].clone()
        elif kind in rowp and "lora_A" in key:
            sh = t.shape[1] // ws
            out[key] = t[:, rank * sh:(rank + 1) * sh].clone()
        else:
            out[key] = t
    return out


@torch.no_grad()
def load_hf_weights_tp(model, model_dir: str, cfg: LlamaConfig,
                       rank: int, ws: int) -> int:
    """Load a REAL local HF-Llama checkpoint into rank `rank`'s TP
    shard (the 13B inference-compare service on actual weights):
    column-parallel q/k/v/gate/up slice output rows, row-parallel
    o/down slice input columns, lm_head is vocab-parallel, embeddings
    and norms replicate. Tied lm_head falls back to the embedding."""
    from ..models.hf_io import _iter_hf_tensors, _map_name
    params = dict(model.named_parameters())
    n = 0
    embed = None
    got_head = False

    def dst(name):
        # row-parallel wrappers insert ".inner" in the module path
        for cand in (name,
                     name.replace("o_proj.", "o_proj.inner.")
                     .replace("down_proj.", "down_proj.inner."),
                     name.replace("lm_head.", "lm_head.inner.")):
            if cand in params:
                return params[cand]
        return None

    for hf_name, t in _iter_hf_tensors(model_dir):
        name = _map_name(hf_name)
        if name is None:
            continue
        p = dst(name)
        if p is None:
            continue
        if any(k in name for k in ("q_proj", "k_proj", "v_proj",
                                   "gate_proj", "up_proj")):
            sh = t.shape[0] // ws
            p.copy_(t[rank * sh:(rank + 1) * sh].to(p.dtype))
        elif any(k in name for k in ("o_proj", "down_proj")):
            sh = t.shape[1] // ws
            p.copy_(t[:, rank * sh:(rank + 1) * sh].to(p.dtype))
        elif name.endswith(("q_bias", "k_bias", "v_bias")):
            # Qwen2 qkv bias: column-parallel like its weight's rows
            sh = t.shape[0] // ws
            p.copy_(t[rank * sh:(rank + 1) * sh].to(p.dtype))
        elif name == "lm_head.weight":
            sh = t.shape[0] // ws
            p.copy_(t[rank * sh:(rank + 1) * sh].to(p.dtype))
            got_head = True
        else:                         # embed, norms: replicated
            p.copy_(t.to(p.dtype))
            if name == "embed_tokens.weight":
                embed = t
        n += 1
    if not got_head and embed is not None:       # tied weights
        p = dst("lm_head.weight")
        sh = embed.shape[0] // ws
        p.copy_(embed[rank * sh:(rank + 1) * sh].to(p.dtype))
        n += 1
    return n
