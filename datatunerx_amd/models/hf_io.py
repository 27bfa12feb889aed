"""Load real HF-format Llama checkpoints and tokenizers from LOCAL
directories (no network) — the reference's AutoTokenizer/
AutoModelForCausalLM.from_pretrained contract
(/root/reference/cmd/tuning/train.py:337, 236-242). A fine-tuning
platform must train and score real models: with this module,
`model_name_or_path` may be a directory containing config.json +
model.safetensors(+index) + tokenizer files, and serve/score evaluates
the actual weights.

Name mapping (HF LlamaForCausalLM -> this repo's LlamaForCausalLM):
  model.embed_tokens.weight            -> embed_tokens.weight
  model.layers.N.self_attn.{q,k,v,o}_proj.weight -> layers.N.self_attn...
  model.layers.N.mlp.{gate,up,down}_proj.weight  -> layers.N.mlp...
  model.layers.N.input_layernorm.weight          -> layers.N.input_layernorm
  model.layers.N.post_attention_layernorm.weight -> ...post_attention_layernorm
  model.norm.weight                    -> norm
  lm_head.weight                       -> lm_head.weight (falls back to
                                          embed_tokens when tied)
No permutation is needed: this repo's RoPE is the HF rotate-half
convention (ops/reference.py:51-61).

Qwen2-family checkpoints load through the same path: `architectures`
containing "Qwen2" sets attention_bias, and
  model.layers.N.self_attn.{q,k,v}_proj.bias -> layers.N.self_attn.{q,k,v}_bias
Qwen3-family sets qk_norm (per-head q/k RMSNorm before RoPE, decoupled
head_dim), with
  model.layers.N.self_attn.{q,k}_norm.weight -> layers.N.self_attn.{q,k}_norm
(Mistral loads as plain Llama geometry; its sliding-window attention
is evaluated as FULL attention — exact for contexts up to the window
size, an over-attention approximation beyond it.) save_hf_model writes the
matching architectures/model_type back, so exports reload in
transformers as the right class.
"""

from __future__ import annotations

import json
import os

import torch

from .llama import LlamaConfig


def is_hf_model_dir(path: str) -> bool:
    return os.path.isdir(path) and os.path.exists(
        os.path.join(path, "config.json"))


def load_hf_config(model_dir: str, **lora_kw) -> LlamaConfig:
    with open(os.path.join(model_dir, "config.json")) as f:
        hc = json.load(f)
    archs = hc.get("architectures") or ["LlamaForCausalLM"]
    if not any("Llama" in a or "Mistral" in a or "Qwen2" in a
               or "Qwen3" in a for a in archs):
        raise ValueError(f"unsupported architecture {archs} in {model_dir}")
    qwen2 = any("Qwen2" in a for a in archs)
    qwen3 = any("Qwen3" in a for a in archs)
    required = ("vocab_size", "hidden_size", "intermediate_size",
                "num_hidden_layers", "num_attention_heads")
    missing = [k for k in required if k not in hc]
    if missing:
        raise ValueError(f"{model_dir}/config.json is missing required "
                         f"fields: {missing}")
    kw = dict(
        vocab_size=hc["vocab_size"],
        hidden_size=hc["hidden_size"],
        intermediate_size=hc["intermediate_size"],
        num_hidden_layers=hc["num_hidden_layers"],
        num_attention_heads=hc["num_attention_heads"],
        num_key_value_heads=hc.get("num_key_value_heads",
                                   hc["num_attention_heads"]),
        max_position_embeddings=hc.get("max_position_embeddings", 4096),
        rms_norm_eps=hc.get("rms_norm_eps", 1e-5),
        rope_theta=hc.get("rope_theta", 10000.0),
        # Qwen2 hardcodes q/k/v bias (no config field in older
        # transformers); Llama-family exposes attention_bias explicitly
        attention_bias=bool(hc.get("attention_bias", qwen2)),
        # Qwen3: per-head q/k RMSNorm before RoPE
        qk_norm=qwen3,
    )
    hd = hc.get("head_dim")
    if hd and hd * kw["num_attention_heads"] != kw["hidden_size"]:
        kw["head_dim_override"] = hd
    kw.update(lora_kw)
    return LlamaConfig(**kw)


def _iter_hf_tensors(model_dir: str):
    """Yield (name, tensor) from model.safetensors, a sharded
    model-*-of-*.safetensors set (via the index), or pytorch_model.bin."""
    from safetensors import safe_open
    idx_path = os.path.join(model_dir, "model.safetensors.index.json")
    single = os.path.join(model_dir, "model.safetensors")
    if os.path.exists(idx_path):
        with open(idx_path) as f:
            index = json.load(f)["weight_map"]
        by_file: dict = {}
        for name, fn in index.items():
            by_file.setdefault(fn, []).append(name)
        for fn, names in by_file.items():
            with safe_open(os.path.join(model_dir, fn), framework="pt") as f:
                for name in names:
                    yield name, f.get_tensor(name)
    elif os.path.exists(single):
        with safe_open(single, framework="pt") as f:
            for name in f.keys():
                yield name, f.get_tensor(name)
    else:
        bin_path = os.path.join(model_dir, "pytorch_model.bin")
        if not os.path.exists(bin_path):
            raise FileNotFoundError(
                f"no model.safetensors(.index.json) or pytorch_model.bin "
                f"in {model_dir}")
        sd = torch.load(bin_path, map_location="cpu", weights_only=True)
        yield from sd.items()


def _map_name(hf_name: str):
    n = hf_name
    if n.startswith("model."):
        n = n[len("model."):]
    if n in ("norm.weight",):
        return "norm"
    if n.endswith("input_layernorm.weight") or \
            n.endswith("post_attention_layernorm.weight"):
        return n[: -len(".weight")]
    if n.startswith("rotary_emb") or "rotary_emb" in n:
        return None                      # computed, not loaded
    for p in ("q", "k", "v"):
        suf = f"self_attn.{p}_proj.bias"
        if n.endswith(suf):              # Qwen2 qkv bias -> {p}_bias
            return n[: -len(suf)] + f"self_attn.{p}_bias"
    if n.endswith(("self_attn.q_norm.weight", "self_attn.k_norm.weight")):
        return n[: -len(".weight")]      # Qwen3 qk-norm
    return n


@torch.no_grad()
def load_hf_weights(model, model_dir: str, strict: bool = True) -> int:
    """Copy HF-Llama weights into `model` (this repo's LlamaForCausalLM).
    Returns the number of tensors loaded. Handles tied lm_head."""
    params = dict(model.named_parameters())
    loaded = set()
    for hf_name, t in _iter_hf_tensors(model_dir):
        name = _map_name(hf_name)
        if name is None:
            continue
        p = params.get(name)
        if p is None:
            if strict and not name.startswith(("lora_",)):
                raise KeyError(f"checkpoint tensor {hf_name!r} has no "
                               f"destination (mapped {name!r})")
            continue
        if p.shape != t.shape:
            raise ValueError(f"{hf_name}: shape {tuple(t.shape)} vs model "
                             f"{tuple(p.shape)}")
        p.copy_(t.to(p.dtype))
        loaded.add(name)
    if "lm_head.weight" not in loaded and "embed_tokens.weight" in loaded:
        params["lm_head.weight"].copy_(params["embed_tokens.weight"])
        loaded.add("lm_head.weight")
    missing = [n for n in params
               if n not in loaded and "lora_" not in n and "_wt" not in n]
    if strict and missing:
        raise KeyError(f"missing weights for: {missing[:8]}"
                       f"{'...' if len(missing) > 8 else ''}")
    return len(loaded)


# ---------------------------------------------------------------- tokenizer
class HFTokenizer:
    """Local-file tokenizer with the ByteTokenizer interface the
    trainer/engine use (encode/decode + special ids). Backends:
    tokenizer.json (HF `tokenizers`) or tokenizer.model
    (SentencePiece)."""

    def __init__(self, backend, kind: str, bos: int, eos: int, pad: int,
                 vocab: int, stop_token_ids=()):
        self._t = backend
        self.kind = kind
        self.bos_token_id = bos
        self.eos_token_id = eos
        self.pad_token_id = pad
        self.vocab_size = vocab
        # llama3-style checkpoints declare eos_token_id as a LIST
        # ([128001, 128008, 128009]); the engine stops on any of these
        self.stop_token_ids = set(stop_token_ids) | {eos}

    @classmethod
    def from_dir(cls, model_dir: str) -> "HFTokenizer":
        bos, eos, pad = 1, 2, None
        stops = set()
        cfg_path = os.path.join(model_dir, "config.json")
        if os.path.exists(cfg_path):
            with open(cfg_path) as f:
                hc = json.load(f)
            bos = hc.get("bos_token_id", bos) or bos
            eos = hc.get("eos_token_id", eos) or eos
            pad = hc.get("pad_token_id", pad)
            if isinstance(eos, (list, tuple)):
                stops = set(int(i) for i in eos)
                eos = int(eos[0])
        tj = os.path.join(model_dir, "tokenizer.json")
        tm = os.path.join(model_dir, "tokenizer.model")
        if os.path.exists(tj):
            from tokenizers import Tokenizer
            t = Tokenizer.from_file(tj)
            vocab = t.get_vocab_size()
            return cls(t, "tokenizers", bos, eos,
                       pad if pad is not None else eos, vocab,
                       stop_token_ids=stops)
        if os.path.exists(tm):
            import sentencepiece as spm
            sp = spm.SentencePieceProcessor(model_file=tm)
            return cls(sp, "sp", sp.bos_id() if sp.bos_id() >= 0 else bos,
                       sp.eos_id() if sp.eos_id() >= 0 else eos,
                       pad if pad is not None else
                       (sp.pad_id() if sp.pad_id() >= 0 else 0),
                       sp.get_piece_size(), stop_token_ids=stops)
        raise FileNotFoundError(
            f"no tokenizer.json or tokenizer.model in {model_dir}")

    def encode(self, text: str, add_special_tokens: bool = False):
        if self.kind == "tokenizers":
            ids = self._t.encode(text, add_special_tokens=False).ids
        else:
            ids = self._t.encode(text, out_type=int)
        if add_special_tokens:
            ids = [self.bos_token_id] + ids
        return ids

    def decode(self, ids):
        drop = self.stop_token_ids | {self.bos_token_id}
        ids = [int(i) for i in ids if int(i) not in drop]
        if self.kind == "tokenizers":
            return self._t.decode(ids)
        return self._t.decode(ids)


def load_tokenizer(name_or_dir: str):
    """HFTokenizer for a local model dir; ByteTokenizer otherwise (the
    builtin names and random-init runs)."""
    if os.path.isdir(name_or_dir) and (
            os.path.exists(os.path.join(name_or_dir, "tokenizer.json")) or
            os.path.exists(os.path.join(name_or_dir, "tokenizer.model"))):
        return HFTokenizer.from_dir(name_or_dir)
    from ..data.dataset import ByteTokenizer
    return ByteTokenizer()


@torch.no_grad()
def save_hf_model(model, out_dir: str):
    """Export this repo's LlamaForCausalLM in HF format (config.json +
    model.safetensors with HF names) so transformers can load the
    trained full-param model — the inverse of load_hf_weights."""
    from safetensors.torch import save_file
    os.makedirs(out_dir, exist_ok=True)
    cfg = model.cfg
    head_dim = getattr(cfg, "head_dim", None) or (
        cfg.hidden_size // cfg.num_attention_heads)
    if getattr(cfg, "qk_norm", False):
        arch, mtype = "Qwen3ForCausalLM", "qwen3"
    elif getattr(cfg, "attention_bias", False):
        arch, mtype = "Qwen2ForCausalLM", "qwen2"
    else:
        arch, mtype = "LlamaForCausalLM", "llama"
    with open(os.path.join(out_dir, "config.json"), "w") as f:
        json.dump({
            "architectures": [arch],
            "model_type": mtype,
            "vocab_size": cfg.vocab_size,
            "hidden_size": cfg.hidden_size,
            "intermediate_size": cfg.intermediate_size,
            "num_hidden_layers": cfg.num_hidden_layers,
            "num_attention_heads": cfg.num_attention_heads,
            "num_key_value_heads": cfg.num_key_value_heads,
            "max_position_embeddings": cfg.max_position_embeddings,
            "rms_norm_eps": cfg.rms_norm_eps,
            "rope_theta": cfg.rope_theta,
            "attention_bias": bool(getattr(cfg, "attention_bias", False)),
            "head_dim": head_dim,
            "torch_dtype": "bfloat16",
            "tie_word_embeddings": False,
        }, f, indent=2)
    sd = {}
    for name, p in model.named_parameters():
        if "lora_" in name:
            continue                    # adapters export separately
        if name == "norm":
            hf = "model.norm.weight"
        elif name.endswith(("self_attn.q_bias", "self_attn.k_bias",
                            "self_attn.v_bias")):
            # inverse of _map_name's Qwen2 bias rule
            hf = "model." + name[:-len("_bias")] + "_proj.bias"
        elif name.endswith(("self_attn.q_norm", "self_attn.k_norm")):
            hf = "model." + name + ".weight"
        elif name == "lm_head.weight":
            hf = "lm_head.weight"
        elif name.endswith(("input_layernorm", "post_attention_layernorm")):
            hf = f"model.{name}.weight"
        else:
            hf = f"model.{name}"
        sd[hf] = p.detach().cpu().contiguous()
    save_file(sd, os.path.join(out_dir, "model.safetensors"))
    return out_dir
