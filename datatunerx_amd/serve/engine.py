"""Inference engine for the inference-compare service.

Replaces the reference's Ray Serve `LlamaDeployment` app (baked into the
checkpoint image — finetunejob_controller.go:378-384):
native PyTorch-ROCm decode through the same fused HIP kernels as
training, with a preallocated KV cache. Single-token decode is
WEIGHT-bandwidth-bound, so throughput scales by request BATCHING
(ragged batched decode, hipGraph-captured batched token step) rather
than by streams; per-stream engine pools cover mixed workloads and the
TP path for 13B is in parallel/tp.py. The serve process is per-job and
torn down after scoring.
"""

from __future__ import annotations

import math
import os
from typing import List, Optional

import torch

from ..data.dataset import ByteTokenizer
from ..data.templates import get_template
from ..models import (GPT2Config, GPT2ForCausalLM, LlamaConfig,
                      LlamaForCausalLM, load_adapter)


class KVCache:
    """BSHD cache ([B, max_s, Hkv, D]) matching the attention kernels'
    native layout — appends along dim 1.

    update() returns (k, v, len_dev): len_dev is None on the eager path
    (the returned tensors are length-exact prefixes) and a device int32
    scalar in graph mode (full-capacity tensors whose live length the
    decode kernel reads on device — hipGraph-replayable)."""

    def __init__(self, B, Hkv, max_s, D, device, dtype,
                 graph_pos=None, graph_len=None):
        self.k = torch.zeros(B, max_s, Hkv, D, device=device, dtype=dtype)
        self.v = torch.zeros(B, max_s, Hkv, D, device=device, dtype=dtype)
        self.cur = 0
        self.graph_pos = graph_pos       # int64 [1]: write index
        self.graph_len = graph_len       # int32 [1]: length incl. new tok

    def update(self, k, v):
        S = k.shape[1]
        if self.graph_pos is not None and S == 1 and k.is_cuda:
            # graph-safe: device-indexed write, device-read length
            self.k.index_copy_(1, self.graph_pos, k)
            self.v.index_copy_(1, self.graph_pos, v)
            return self.k, self.v, self.graph_len
        self.k[:, self.cur:self.cur + S] = k
        self.v[:, self.cur:self.cur + S] = v
        self.cur += S
        kk, vv = self.k[:, :self.cur], self.v[:, :self.cur]
        if self.k.shape[0] > 1 or not k.is_cuda:
            # batch>1 / CPU paths want contiguous; the GPU decode kernel
            # takes the dense prefix view directly (no per-token copy)
            kk, vv = kk.contiguous(), vv.contiguous()
        return kk, vv, None


class _BatchedGraphedDecoder:
    """hipGraph-captured BATCHED decode step (fixed batch MAXB, padded
    rows): one replay per token for the whole batch, greedy argmax
    in-graph, per-row position/length counters device-resident. Built
    once per engine (under the capture lock) and reused; requests pad
    to MAXB with dummy rows."""

    def __init__(self, model, cfg, device, B, max_s):
        self.B, self.max_s = B, max_s
        dtype = next(model.parameters()).dtype
        self.pos64 = torch.zeros(B, dtype=torch.long, device=device)
        self.pos32 = torch.zeros(B, dtype=torch.int32, device=device)
        self.len32 = torch.zeros(B, dtype=torch.int32, device=device)
        self.caches = [
            BatchedKVCache(B, cfg.num_key_value_heads, max_s,
                           cfg.head_dim, device, dtype,
                           pos64=self.pos64, len32=self.len32)
            for _ in range(cfg.num_hidden_layers)]
        self.input = torch.ones(B, 1, dtype=torch.long, device=device)
        self.model = model
        with torch.no_grad():
            for _ in range(3):
                lg = model(self.input, pos_dev=self.pos32,
                           kv_caches=self.caches)
                lg[:, -1].argmax(-1)
            torch.cuda.synchronize()
            self.graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.graph):
                lg = model(self.input, pos_dev=self.pos32,
                           kv_caches=self.caches)
                self.out = lg[:, -1].argmax(-1)

    def reset(self):
        self.pos64.zero_()
        self.pos32.zero_()
        self.len32.zero_()

    def set_state(self, lens):
        pos = torch.tensor(lens, dtype=torch.long,
                           device=self.pos64.device)
        self.pos64.copy_(pos)
        self.pos32.copy_(pos.to(torch.int32))
        self.len32.copy_((pos + 1).to(torch.int32))

    def step(self, cur):
        """cur: [B,1] int64 device. Returns next tokens [B] (device)."""
        self.input.copy_(cur)
        self.graph.replay()
        self.pos64 += 1
        self.pos32 += 1
        self.len32 += 1
        return self.out


class BatchedKVCache:
    """Ragged batched decode cache: rows prefill individually (row_view
    writes through a dim-0 slice), then decode appends ONE token per row
    at per-row positions (pos64 [B] int64) and attends per-row lengths
    (len32 [B] int32 — the decode kernel / CPU ref read them per b).
    Batched decode shares one weight stream across the batch — the
    single-stream decode is WEIGHT-bandwidth-bound, so this is the
    scaling axis for serving throughput."""

    def __init__(self, B, Hkv, max_s, D, device, dtype,
                 pos64=None, len32=None):
        self.k = torch.zeros(B, max_s, Hkv, D, device=device, dtype=dtype)
        self.v = torch.zeros(B, max_s, Hkv, D, device=device, dtype=dtype)
        # pos/len may be SHARED across layers (the graphed decoder
        # advances one set of counters for all layer caches)
        self.pos64 = (pos64 if pos64 is not None else
                      torch.zeros(B, dtype=torch.long, device=device))
        self.len32 = (len32 if len32 is not None else
                      torch.zeros(B, dtype=torch.int32, device=device))
        self._rows = torch.arange(B, device=device)

    class _RowView:
        def __init__(self, parent, i):
            self.parent, self.i = parent, i
            self.cur = 0

        def update(self, k, v):
            S = k.shape[1]
            self.parent.k[self.i, self.cur:self.cur + S] = k[0]
            self.parent.v[self.i, self.cur:self.cur + S] = v[0]
            self.cur += S
            return (self.parent.k[self.i:self.i + 1, :self.cur],
                    self.parent.v[self.i:self.i + 1, :self.cur], None)

    def row_view(self, i):
        return BatchedKVCache._RowView(self, i)

    def update(self, k, v):
        if k.shape[1] > 1:
            # batched RIGHT-PADDED prefill: write all rows up to the
            # padded length; per-row len32 (set by the caller) bounds
            # every later read, and decode overwrites the first pad
            # slot, so pad K/V beyond a row's real length is never
            # consumed. Returns len_dev=None: prefill attention runs
            # plain causal over the padded batch (pads sit AFTER real
            # tokens, so no real query attends one).
            S = k.shape[1]
            self.k[:, :S] = k
            self.v[:, :S] = v
            return k, v, None
        # batched decode append: k/v [B, 1, Hkv, D] at per-row positions
        self.k[self._rows, self.pos64] = k[:, 0]
        self.v[self._rows, self.pos64] = v[:, 0]
        return self.k, self.v, self.len32


def builtin_config(name: str, **lora_kw):
    """Resolve a builtin model name to ("llama"|"gpt2", config) without
    allocating weights — the single registry the engine, TP server and
    tests share. Raises ValueError on unknown names."""
    if name in ("llama2-7b", "llama-2-7b"):
        return "llama", LlamaConfig.llama2_7b(**lora_kw)
    if name in ("llama2-13b", "llama-2-13b"):
        return "llama", LlamaConfig.llama2_13b(**lora_kw)
    if name in ("llama3-8b", "llama-3-8b"):
        return "llama", LlamaConfig.llama3_8b(**lora_kw)
    if name == "llama-tiny":
        return "llama", LlamaConfig.tiny(**lora_kw)
    if name == "llama-mini":
        return "llama", LlamaConfig.mini(**lora_kw)
    if name in ("gpt2-small", "gpt2"):
        return "gpt2", GPT2Config.small(**lora_kw)
    if name == "gpt2-tiny":
        return "gpt2", GPT2Config.tiny(**lora_kw)
    raise ValueError(f"unknown model {name!r}")


def build_model(name: str, device, adapter_dir: Optional[str] = None,
                lora_kw: Optional[dict] = None):
    dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
    lora_kw = dict(lora_kw or {})
    if adapter_dir and not lora_kw:
        # adapt the LoRA geometry to the checkpoint's adapter_config.json
        import json
        import os
        cfg_path = os.path.join(adapter_dir, "adapter_config.json")
        if os.path.exists(cfg_path):
            with open(cfg_path) as f:
                ac = json.load(f)
            lora_kw = {"lora_r": int(ac.get("r", 8)),
                       "lora_alpha": float(ac.get("lora_alpha", 32)),
                       "lora_targets": tuple(ac.get("target_modules") or
                                             ("q_proj", "v_proj"))}
    from ..models.hf_io import (is_hf_model_dir, load_hf_config,
                                load_hf_weights)
    hf_dir = is_hf_model_dir(name)
    with torch.device(device):
        if hf_dir:
            model = LlamaForCausalLM(load_hf_config(name, **lora_kw),
                                     lora=bool(adapter_dir), dtype=dtype)
        else:
            family, cfg = builtin_config(name, **lora_kw)
            if family == "llama":
                model = LlamaForCausalLM(cfg, dtype=dtype)
            else:
                model = GPT2ForCausalLM(cfg, dtype=dtype)
    if hf_dir:
        load_hf_weights(model, name)   # real local weights
    else:
        model.init_random()      # builtin names: random-init (no network)
    if adapter_dir:
        load_adapter(model, adapter_dir)
    model.eval()
    return model


class _GraphedDecoder:
    """hipGraph-captured decode step (torch.cuda.CUDAGraph): the whole
    32-layer token forward replays as one graph — position, cache length
    and the KV write index live in device tensors so the replay needs no
    re-capture as the sequence grows. Built once per engine at fixed
    cache capacity; counters reset per request."""

    def __init__(self, model, cfg, device, max_s):
        self.model, self.max_s = model, max_s
        dtype = next(model.parameters()).dtype
        self.pos64 = torch.zeros(1, dtype=torch.long, device=device)
        self.pos32 = torch.zeros(1, dtype=torch.int32, device=device)
        self.len32 = torch.ones(1, dtype=torch.int32, device=device)
        self.caches = [
            KVCache(1, cfg.num_key_value_heads, max_s, cfg.head_dim,
                    device, dtype, graph_pos=self.pos64,
                    graph_len=self.len32)
            for _ in range(cfg.num_hidden_layers)]
        self.input_id = torch.zeros(1, 1, dtype=torch.long, device=device)
        # warmup (allocator + kernels), then capture
        with torch.no_grad():
            for _ in range(3):
                self.model(self.input_id, pos_dev=self.pos32,
                           kv_caches=self.caches)
            torch.cuda.synchronize()
            self.graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.graph):
                self.logits = self.model(self.input_id,
                                         pos_dev=self.pos32,
                                         kv_caches=self.caches)

    def reset(self):
        self.pos64.zero_()
        self.pos32.zero_()
        self.len32.fill_(1)
        for c in self.caches:
            c.cur = 0

    def after_prefill(self, n: int):
        """counters for the first graphed token (written at index n)."""
        self.pos64.fill_(n)
        self.pos32.fill_(n)
        self.len32.fill_(n + 1)
        for c in self.caches:
            c.cur = n

    def step(self, token: int):
        self.input_id[0, 0] = token
        self.graph.replay()
        self.pos64 += 1
        self.pos32 += 1
        self.len32 += 1
        return self.logits


class InferenceEngine:
    MAX_BATCH = int(os.environ.get("DTX_SERVE_MAX_BATCH", "8"))

    """One generation context over a (shared, read-only) model. For
    concurrent serving, build several engines over the SAME model
    (serve/server.py EnginePool): each holds its own KV caches, hip
    graph and HIP stream, so batch-1 decodes overlap on the GPU instead
    of queueing on one engine lock (decode is latency-bound, not
    occupancy-bound)."""

    def __init__(self, model, tokenizer=None, template: str = "llama2",
                 device=None, graph_decode: bool = True,
                 own_stream: bool = False):
        self.model = model
        self.tok = tokenizer or ByteTokenizer()
        self.template = template
        self.device = device or next(model.parameters()).device
        self.is_llama = isinstance(model, LlamaForCausalLM)
        import torch.distributed as dist
        self._graphed = None
        self._bgraphed = None
        self._stream = (torch.cuda.Stream(device=self.device)
                        if own_stream and self.device.type == "cuda"
                        else None)
        self._graph_ok = (graph_decode and self.is_llama and
                          self.device.type == "cuda" and
                          not dist.is_initialized() and
                          os.environ.get("DTX_NO_GRAPH") != "1")
        # stop set: tokenizer eos (may be a LIST in llama3-style
        # config.json), plus any template stop_words that the tokenizer
        # encodes to a single special id (llama3's <|eot_id|> ends a
        # turn but is distinct from eos)
        eos = getattr(self.tok, "eos_token_id", None)
        self._stop_ids = set(eos) if isinstance(eos, (list, tuple)) \
            else ({eos} if eos is not None else set())
        extra = getattr(self.tok, "stop_token_ids", None)
        if extra:
            self._stop_ids.update(int(i) for i in extra)
        try:
            tmpl = get_template(template)
            for w in tmpl.stop_words:
                ids = self.tok.encode(w, add_special_tokens=False) \
                    if hasattr(self.tok, "encode") else []
                if len(ids) == 1:
                    self._stop_ids.add(int(ids[0]))
        except (KeyError, TypeError):
            pass

    def _is_stop(self, tok_id) -> bool:
        return int(tok_id) in self._stop_ids

    def _stream_ctx(self):
        import contextlib
        if self._stream is None:
            return contextlib.nullcontext()
        return torch.cuda.stream(self._stream)

    # hip graph CAPTURE must never run concurrently with another
    # engine's capture or in-flight kernels (capture mode is
    # context-global); all captures serialize on this lock. Replays of
    # distinct graphs on distinct streams are concurrency-safe.
    _capture_lock = __import__("threading").Lock()

    def _get_graphed(self):
        if self._graphed is None and self._graph_ok:
            try:
                with InferenceEngine._capture_lock, self._stream_ctx():
                    cfg = self.model.cfg
                    self._graphed = _GraphedDecoder(
                        self.model, cfg, self.device,
                        cfg.max_position_embeddings)
            except Exception:
                self._graph_ok = False
        return self._graphed

    def warmup(self):
        """Build the graphed decoder + run one tiny generation NOW (a
        pool builds engines sequentially before serving concurrent
        requests, so no capture happens mid-traffic)."""
        with self._stream_ctx():
            self._get_graphed()
            list(self.generate([1], max_new_tokens=2))
        if self.device.type == "cuda":
            torch.cuda.synchronize()

    # ------------------------------------------------------------ chat
    def chat(self, messages: List[dict], max_tokens: int = 64,
             temperature: float = 0.0, top_p: float = 1.0) -> str:
        """messages: [{role, content}] -> completion text."""
        with self._stream_ctx():
            return self._chat(messages, max_tokens, temperature, top_p)

    def _chat(self, messages: List[dict], max_tokens: int = 64,
              temperature: float = 0.0, top_p: float = 1.0) -> str:
        system = ""
        history = []
        query = ""
        pending_user = None
        for m in messages:
            if m["role"] == "system":
                system = m["content"]
            elif m["role"] == "user":
                pending_user = m["content"]
            elif m["role"] == "assistant" and pending_user is not None:
                history.append((pending_user, m["content"]))
                pending_user = None
        query = pending_user or ""
        t = get_template(self.template)
        src, _ = t.encode_oneturn(self.tok, query, "", history, system)
        ids = self.generate(src, max_tokens, temperature, top_p)
        return self.tok.decode(ids)

    def chat_stream(self, messages: List[dict], max_tokens: int = 64,
                    temperature: float = 0.0, top_p: float = 1.0):
        for delta in self._chat_stream_inner(messages, max_tokens,
                                             temperature, top_p):
            yield delta

    def _chat_stream_inner(self, messages, max_tokens, temperature,
                           top_p):
        """Like chat() but yields text DELTAS as tokens decode (the
        serving endpoint streams these as SSE chunks). Token-identical
        to chat(): same generate loop, so TP followers running chat()
        stay in collective lockstep with a streaming front rank."""
        system, history, pending_user = "", [], None
        for m in messages:
            if m["role"] == "system":
                system = m["content"]
            elif m["role"] == "user":
                pending_user = m["content"]
            elif m["role"] == "assistant" and pending_user is not None:
                history.append((pending_user, m["content"]))
                pending_user = None
        t = get_template(self.template)
        src, _ = t.encode_oneturn(self.tok, pending_user or "", "",
                                  history, system)
        ids, sent = [], ""
        for tok_id in self.generate_stream(src, max_tokens, temperature,
                                           top_p):
            ids.append(tok_id)
            full = self.tok.decode(ids)
            # hold back trailing U+FFFD: a multi-byte UTF-8 char split
            # across tokens decodes as a replacement char until the
            # rest arrives; emitting it early would make the
            # concatenated stream differ from chat().
            stable = full
            while stable.endswith("�"):
                stable = stable[:-1]
            if len(stable) > len(sent):
                delta, sent = stable[len(sent):], stable
                yield delta
        full = self.tok.decode(ids)
        if len(full) > len(sent):
            yield full[len(sent):]      # flush any held-back tail

    @torch.no_grad()
    def generate(self, prompt_ids: List[int], max_new_tokens: int = 64,
                 temperature: float = 0.0, top_p: float = 1.0) -> List[int]:
        return list(self.generate_stream(prompt_ids, max_new_tokens,
                                         temperature, top_p))

    @torch.no_grad()
    def generate_stream(self, prompt_ids: List[int],
                        max_new_tokens: int = 64, temperature: float = 0.0,
                        top_p: float = 1.0):
        with self._stream_ctx():
            yield from self._generate_stream(prompt_ids, max_new_tokens,
                                             temperature, top_p)

    @torch.no_grad()
    def _generate_stream(self, prompt_ids: List[int],
                         max_new_tokens: int = 64,
                         temperature: float = 0.0, top_p: float = 1.0):
        if not prompt_ids:
            raise ValueError("empty prompt")
        limit = getattr(self.model.cfg, "max_position_embeddings", 1 << 30)
        if len(prompt_ids) >= limit:
            raise ValueError(
                f"prompt length {len(prompt_ids)} exceeds the model "
                f"context ({limit} positions)")
        ids = torch.tensor([prompt_ids], dtype=torch.long,
                           device=self.device)
        n_out = 0
        gd = self._get_graphed() if self.is_llama else None
        if gd is not None and len(prompt_ids) + max_new_tokens + 1 < \
                gd.max_s:
            # hipGraph-replayed decode: prefill eagerly into the graph's
            # caches, then one graph replay per token
            gd.reset()
            logits = self.model(ids, pos0=0, kv_caches=gd.caches)
            gd.after_prefill(len(prompt_ids))
            nxt = self._sample(logits[0, -1], temperature, top_p)
            for _ in range(max_new_tokens):
                if self._is_stop(nxt):
                    break
                yield nxt
                n_out += 1
                if len(prompt_ids) + n_out + 1 >= gd.max_s:
                    break
                logits = gd.step(nxt)
                nxt = self._sample(logits[0, -1], temperature, top_p)
            return
        if self.is_llama:
            cfg = self.model.cfg
            max_s = min(cfg.max_position_embeddings,
                        len(prompt_ids) + max_new_tokens + 8)
            caches = [KVCache(1, cfg.num_key_value_heads, max_s,
                              cfg.head_dim, self.device,
                              next(self.model.parameters()).dtype)
                      for _ in range(cfg.num_hidden_layers)]
            pos = 0
            cur = ids
            for _ in range(max_new_tokens):
                logits = self.model(cur, pos0=pos, kv_caches=caches)
                pos += cur.shape[1]
                if pos >= max_s:
                    break
                nxt = self._sample(logits[0, -1], temperature, top_p)
                if self._is_stop(nxt):
                    break
                yield nxt
                cur = torch.tensor([[nxt]], dtype=torch.long,
                                   device=self.device)
        else:
            seq = list(prompt_ids)
            for _ in range(max_new_tokens):
                cur = torch.tensor([seq], dtype=torch.long,
                                   device=self.device)
                logits = self.model(cur)
                nxt = self._sample(logits[0, -1], temperature, top_p)
                if self._is_stop(nxt):
                    break
                yield nxt
                seq.append(nxt)

    def _sample(self, logits: torch.Tensor, temperature: float,
                top_p: float) -> int:
        import torch.distributed as dist
        tp = dist.is_initialized() and dist.get_world_size() > 1
        comm_dev = (self.device if tp and dist.get_backend() == "nccl"
                    else torch.device("cpu"))
        if tp and dist.get_rank() != 0:
            # follower ranks take rank 0's choice (RNG is per-process)
            t = torch.zeros(1, dtype=torch.long, device=comm_dev)
            dist.broadcast(t, src=0)
            return int(t)
        if temperature <= 0.0:
            nxt = int(logits.argmax())
        else:
            probs = torch.softmax(logits.float() / temperature, dim=-1)
            if top_p < 1.0:
                sp, si = probs.sort(descending=True)
                cum = sp.cumsum(0)
                keep = cum <= top_p
                keep[0] = True
                probs = torch.zeros_like(probs).scatter_(0, si[keep],
                                                         sp[keep])
                probs /= probs.sum()
            nxt = int(torch.multinomial(probs, 1))
        if tp:
            dist.broadcast(torch.tensor([nxt], dtype=torch.long,
                                        device=comm_dev), src=0)
        return nxt

    # ------------------------------------------------- batched decode
    @torch.no_grad()
    def generate_batch(self, prompts: List[List[int]],
                       max_new_tokens: int = 64,
                       temperature: float = 0.0,
                       top_p: float = 1.0) -> List[List[int]]:
        """Ragged batched generation: per-row prefill, then ONE decode
        step per token for the whole batch (the weight stream is read
        once per step instead of once per request)."""
        with self._stream_ctx():
            return self._generate_batch(prompts, max_new_tokens,
                                        temperature, top_p)

    def _get_batched_graphed(self, max_s):
        if self._bgraphed is None and self._graph_ok:
            try:
                with InferenceEngine._capture_lock, self._stream_ctx():
                    self._bgraphed = _BatchedGraphedDecoder(
                        self.model, self.model.cfg, self.device,
                        InferenceEngine.MAX_BATCH, max_s)
            except Exception:
                self._bgraphed = False
        return self._bgraphed or None

    @torch.no_grad()
    def _generate_batch(self, prompts, max_new_tokens, temperature,
                        top_p, budgets=None):
        outs = [[] for _ in prompts]
        for emitted in self._generate_batch_steps(
                prompts, max_new_tokens, temperature, top_p, budgets):
            for i, t in emitted:
                outs[i].append(t)
        return outs

    @torch.no_grad()
    def _generate_batch_steps(self, prompts, max_new_tokens, temperature,
                              top_p, budgets=None):
        """Yields, per decode step, the [(row, token)] emissions.
        budgets: optional per-row max_tokens (rows stop emitting — and
        count as done — at their own budget)."""
        assert self.is_llama, "batched decode is the Llama path"
        cfg = self.model.cfg
        greedy = not (temperature and temperature > 0)
        lens0 = [len(p) for p in prompts]
        if not prompts or min(lens0) == 0:
            raise ValueError("empty prompt in batch")
        gd = None
        if greedy and self.device.type == "cuda" and \
                len(prompts) <= InferenceEngine.MAX_BATCH:
            cap = min(cfg.max_position_embeddings,
                      int(os.environ.get("DTX_SERVE_MAXS", "2048")))
            if max(lens0) + max_new_tokens + 8 < cap:
                gd = self._get_batched_graphed(cap)
        if gd is not None:
            # pad the batch to the graph's fixed MAXB with dummy rows
            prompts = list(prompts) + \
                [[self.tok.bos_token_id]] * (gd.B - len(prompts))
        B = len(prompts)
        lens = [len(p) for p in prompts]
        max_s = gd.max_s if gd is not None else min(
            cfg.max_position_embeddings,
            max(lens) + max_new_tokens + 8)
        dtype = next(self.model.parameters()).dtype
        if gd is not None:
            # no cache zeroing needed: prefill overwrites rows up to
            # len and attention never reads past len32
            gd.reset()
            caches = gd.caches
        else:
            caches = [BatchedKVCache(B, cfg.num_key_value_heads, max_s,
                                     cfg.head_dim, self.device, dtype)
                      for _ in range(cfg.num_hidden_layers)]
        nxt = [0] * B
        lmax = max(lens)
        if B > 1 and self.device.type == "cuda":
            # ONE right-padded batched prefill (pads never attended:
            # causal + pads-after-real; len32 bounds decode reads)
            padded = torch.zeros(B, lmax, dtype=torch.long,
                                 device=self.device)
            for i, ids in enumerate(prompts):
                padded[i, :len(ids)] = torch.tensor(ids,
                                                    dtype=torch.long)
            logits = self.model(padded, pos0=0, kv_caches=caches)
            for i in range(B):
                nxt[i] = self._sample(logits[i, lens[i] - 1],
                                      temperature, top_p)
        else:
            for i, ids in enumerate(prompts):
                row = [c.row_view(i) for c in caches]
                t = torch.tensor([ids], dtype=torch.long,
                                 device=self.device)
                logits = self.model(t, pos0=0, kv_caches=row)
                nxt[i] = self._sample(logits[0, -1], temperature, top_p)
        pos = torch.tensor(lens, dtype=torch.long, device=self.device)
        if gd is not None:
            gd.set_state(lens)
            pos32 = gd.pos32
        else:
            for c in caches:
                c.pos64.copy_(pos)
                c.len32.copy_((pos + 1).to(torch.int32))
            pos32 = pos.to(torch.int32)
        n_live = len(lens0)
        if budgets is None:
            budgets = [max_new_tokens] * n_live
        count = [0] * n_live
        done = [False] * B
        first = []
        for i in range(B):
            if self._is_stop(nxt[i]):
                done[i] = True
            elif i < n_live:
                if count[i] < budgets[i]:
                    first.append((i, nxt[i]))
                    count[i] += 1
                if count[i] >= budgets[i]:
                    done[i] = True
        yield first
        cur = torch.tensor(nxt, dtype=torch.long,
                           device=self.device).view(B, 1)
        p32 = pos32
        cap = max_s - max(lens) - 2
        for _ in range(min(max_new_tokens - 1, cap)):
            if all(done[:n_live]):
                break
            if gd is not None:
                out = gd.step(cur)
                toks = out.tolist()
                cur = out.view(B, 1)
            else:
                logits = self.model(cur, kv_caches=caches, pos_dev=p32)
                for c in caches:
                    c.pos64 += 1
                    c.len32 += 1
                p32 += 1
                if temperature and temperature > 0:
                    toks = [self._sample(logits[i, -1], temperature,
                                         top_p) for i in range(B)]
                else:
                    toks = logits[:, -1].argmax(-1).tolist()
                cur = torch.tensor(toks, dtype=torch.long,
                                   device=self.device).view(B, 1)
            emitted = []
            for i, t in enumerate(toks):
                if done[i]:
                    continue
                if self._is_stop(t):
                    done[i] = True
                elif i < n_live:
                    emitted.append((i, t))
                    count[i] += 1
                    if count[i] >= budgets[i]:
                        done[i] = True
            yield emitted

    def _encode_requests(self, requests):
        t = get_template(self.template)
        prompts = []
        for r in requests:
            system, history, pending = "", [], None
            for m in r.get("messages", []):
                if m["role"] == "system":
                    system = m["content"]
                elif m["role"] == "user":
                    pending = m["content"]
                elif m["role"] == "assistant" and pending is not None:
                    history.append((pending, m["content"]))
                    pending = None
            src, _ = t.encode_oneturn(self.tok, pending or "", "",
                                      history, system)
            prompts.append(src)
        budgets = [int(r.get("max_tokens", 64)) for r in requests]
        temp = float(requests[0].get("temperature", 0.0))
        top_p = float(requests[0].get("top_p", 1.0))
        for r in requests[1:]:
            if (float(r.get("temperature", 0.0)),
                    float(r.get("top_p", 1.0))) != (temp, top_p):
                # one batched generation samples with ONE setting; the
                # BatchingFront groups requests by it before batching
                raise ValueError("batched requests must share "
                                 "temperature/top_p")
        return prompts, budgets, temp, top_p

    def chat_batch(self, requests: List[dict]) -> List[str]:
        """requests: [{messages, max_tokens, temperature, top_p}] ->
        completion texts (one batched generation)."""
        prompts, budgets, temp, top_p = self._encode_requests(requests)
        with self._stream_ctx():
            outs = self._generate_batch(prompts, max(budgets), temp,
                                        top_p, budgets=budgets)
        return [self.tok.decode(o) for o in outs]

    def chat_batch_stream(self, requests: List[dict]):
        """Batched STREAMING generation: yields (row_index, text_delta)
        as tokens decode, then (row_index, None) when a row finishes.
        Deltas per row concatenate to exactly chat_batch()'s text (the
        same U+FFFD holdback as chat_stream)."""
        prompts, budgets, temp, top_p = self._encode_requests(requests)
        n = len(requests)
        ids = [[] for _ in range(n)]
        sent = [""] * n
        live = [True] * n
        with self._stream_ctx():
            for emitted in self._generate_batch_steps(
                    prompts, max(budgets), temp, top_p, budgets=budgets):
                step_rows = set()
                for i, t in emitted:
                    ids[i].append(t)
                    step_rows.add(i)
                for i in step_rows:
                    full = self.tok.decode(ids[i])
                    stable = full
                    while stable.endswith("\ufffd"):
                        stable = stable[:-1]
                    if len(stable) > len(sent[i]):
                        delta = stable[len(sent[i]):]
                        sent[i] = stable
                        yield i, delta
        for i in range(n):
            if live[i]:
                full = self.tok.decode(ids[i])
                if len(full) > len(sent[i]):
                    yield i, full[len(sent[i]):]
                yield i, None

    # ----------------------------------------------------------- score
    @torch.no_grad()
    def perplexity(self, texts: List[str]) -> float:
        with self._stream_ctx():
            return self._perplexity(texts)

    def _perplexity(self, texts: List[str]) -> float:
        """Mean perplexity over texts (built-in Scoring metric)."""
        losses = []
        for t in texts:
            ids = [self.tok.bos_token_id] + self.tok.encode(t)
            ids = ids[:512]
            if len(ids) < 2:
                continue
            x = torch.tensor([ids], dtype=torch.long, device=self.device)
            loss = self.model(x, labels=x)
            losses.append(float(loss))
        if not losses:
            return float("inf")
        mean = sum(losses) / len(losses)
        return math.exp(min(mean, 30.0))
