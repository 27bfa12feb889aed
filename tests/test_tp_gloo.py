"""Tensor-parallel inference tests on CPU (gloo, world_size=2):
the TP-sharded Llama must produce the same logits as the single-process
model with the same seed (SURVEY.md §7 step 7 — the inference-compare
service's TP path, exercised shape-for-shape on gloo)."""

import json
import os
import subprocess
import sys

import pytest
import torch

from conftest import free_port

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import json, os, sys
import torch
import torch.distributed as dist
sys.path.insert(0, os.environ["DTX_ROOT"])
from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM
from datatunerx_amd.parallel.ddp import init_distributed
from datatunerx_amd.parallel.tp import build_tp_llama

rank, world, local, device = init_distributed(backend="gloo")
cfg = LlamaConfig.tiny()
tp = build_tp_llama(cfg, rank, world, lora=True, dtype=torch.float32,
                    device=device, seed=5)
torch.manual_seed(11)
ids = torch.randint(0, cfg.vocab_size, (2, 24))
dist.broadcast(ids, src=0)
with torch.no_grad():
    logits_tp = tp(ids)
if rank == 0:
    full = LlamaForCausalLM(cfg, lora=True,
                            dtype=torch.float32).init_random(seed=5)
    with torch.no_grad():
        logits_full = full(ids)
    err = (logits_tp - logits_full).abs().max().item()
    ref = logits_full.abs().max().item()
    with open(os.environ["DTX_OUT"] + "/out.json", "w") as f:
        json.dump({"err": err, "ref": ref}, f)
dist.destroy_process_group()
"""


def test_tp_matches_single(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    _port = str(free_port())
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update({"RANK": str(rank), "WORLD_SIZE": "2",
                    "LOCAL_RANK": str(rank),
                    "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": _port,
                    "DTX_ROOT": ROOT, "DTX_OUT": str(tmp_path)})
        procs.append(subprocess.Popen([sys.executable, str(script)],
                                      env=env))
    for p in procs:
        assert p.wait(timeout=300) == 0
    out = json.load(open(tmp_path / "out.json"))
    assert out["err"] < 1e-4 * max(1.0, out["ref"])


def test_tp_server_end_to_end(tmp_path):
    """2-rank TP server on gloo answers /chat/completions and /v1/score
    with the same completion as the single-process engine."""
    import time
    import urllib.request

    port = 18973
    _port = str(free_port())
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update({"RANK": str(rank), "WORLD_SIZE": "2",
                    "LOCAL_RANK": str(rank),
                    "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": _port,
                    "PYTHONPATH": ROOT})
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "datatunerx_amd.serve.server",
             "--model", "llama-tiny", "--port", str(port),
             "--template", "vanilla"], env=env))
    try:
        body = json.dumps({"messages": [{"role": "user", "content": "hi"}],
                           "max_tokens": 8}).encode()
        deadline = time.time() + 120
        resp = None
        while time.time() < deadline:
            try:
                req = urllib.request.Request(
                    f"http://127.0.0.1:{port}/chat/completions", body,
                    {"Content-Type": "application/json"})
                resp = json.load(urllib.request.urlopen(req, timeout=10))
                break
            except Exception:
                if any(p.poll() is not None for p in procs):
                    raise AssertionError("TP server rank died early")
                time.sleep(1.0)
        assert resp is not None, "server never came up"
        tp_text = resp["choices"][0]["message"]["content"]
        req = urllib.request.Request(
            f"http://127.0.0.1:{port}/v1/score",
            json.dumps({"texts": ["hello world"]}).encode(),
            {"Content-Type": "application/json"})
        score = json.load(urllib.request.urlopen(req, timeout=30))
        assert score["perplexity"] > 0
    finally:
        for p in procs:
            p.kill()
    # single-process engine with the same seed must emit the same text
    import torch as _t

    sys.path.insert(0, ROOT)
    from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM
    from datatunerx_amd.serve.engine import InferenceEngine
    model = LlamaForCausalLM(LlamaConfig.tiny(), lora=False,
                             dtype=_t.float32).init_random(seed=0)
    model.eval()
    eng = InferenceEngine(model, template="vanilla",
                          device=_t.device("cpu"))
    ref_text = eng.chat([{"role": "user", "content": "hi"}], 8)
    assert tp_text == ref_text


def test_shard_adapter_state_roundtrip():
    """Sharding a PEFT adapter splits B rows (col-parallel) and A cols
    (row-parallel) and leaves the rest alone."""
    from datatunerx_amd.parallel.tp import shard_adapter_state
    from datatunerx_amd.models import LlamaConfig
    cfg = LlamaConfig.tiny()
    sd = {
        "base_model.model.layers.0.self_attn.q_proj.lora_A.weight":
            torch.randn(8, cfg.hidden_size),
        "base_model.model.layers.0.self_attn.q_proj.lora_B.weight":
            torch.randn(cfg.hidden_size, 8),
        "base_model.model.layers.0.self_attn.o_proj.lora_A.weight":
            torch.randn(8, cfg.hidden_size),
        "base_model.model.layers.0.self_attn.o_proj.lora_B.weight":
            torch.randn(cfg.hidden_size, 8),
    }
    s0 = shard_adapter_state(sd, cfg, 0, 2)
    s1 = shard_adapter_state(sd, cfg, 1, 2)
    qb = "base_model.model.layers.0.self_attn.q_proj.lora_B.weight"
    oa = "base_model.model.layers.0.self_attn.o_proj.lora_A.weight"
    qa = "base_model.model.layers.0.self_attn.q_proj.lora_A.weight"
    assert s0[qb].shape[0] == cfg.hidden_size // 2
    assert torch.equal(torch.cat([s0[qb], s1[qb]], 0), sd[qb])
    assert torch.equal(torch.cat([s0[oa], s1[oa]], 1), sd[oa])
    assert torch.equal(s0[qa], sd[qa])


def test_tp_adapter_loading(tmp_path):
    """A full HF-PEFT adapter loads into TP shards and reproduces the
    single-process adapter-model logits (2 ranks, gloo)."""
    worker = r"""
import json, os, sys
import torch
import torch.distributed as dist
sys.path.insert(0, os.environ["DTX_ROOT"])
from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM
from datatunerx_amd.models.lora import save_adapter, load_adapter
from datatunerx_amd.parallel.ddp import init_distributed
from datatunerx_amd.parallel.tp import build_tp_llama, load_adapter_tp

rank, world, local, device = init_distributed(backend="gloo")
cfg = LlamaConfig.tiny(lora_targets=("q_proj", "v_proj", "o_proj"))
out = os.environ["DTX_OUT"]
if rank == 0:
    full = LlamaForCausalLM(cfg, lora=True,
                            dtype=torch.float32).init_random(seed=5)
    torch.manual_seed(3)
    for n, p in full.named_parameters():
        if "lora" in n:
            with torch.no_grad():
                p.add_(torch.randn_like(p) * 0.05)
    save_adapter(full, out + "/ad", r=cfg.lora_r, alpha=cfg.lora_alpha,
                 dropout=0.0, target_modules=list(cfg.lora_targets))
dist.barrier()
tp = build_tp_llama(cfg, rank, world, lora=True, dtype=torch.float32,
                    device=device, seed=5)
n = load_adapter_tp(tp, out + "/ad", cfg, rank, world)
assert n > 0, "no adapter tensors loaded"
tp.eval()
torch.manual_seed(11)
ids = torch.randint(0, cfg.vocab_size, (1, 16))
dist.broadcast(ids, src=0)
with torch.no_grad():
    lt = tp(ids)
if rank == 0:
    full2 = LlamaForCausalLM(cfg, lora=True,
                             dtype=torch.float32).init_random(seed=5)
    load_adapter(full2, out + "/ad")
    full2.eval()
    with torch.no_grad():
        lf = full2(ids)
    err = (lt - lf).abs().max().item()
    with open(out + "/ad_out.json", "w") as f:
        json.dump({"err": err, "ref": lf.abs().max().item()}, f)
dist.destroy_process_group()
"""
    script = tmp_path / "w2.py"
    script.write_text(worker)
    _port = str(free_port())
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update({"RANK": str(rank), "WORLD_SIZE": "2",
                    "LOCAL_RANK": str(rank),
                    "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": _port,
                    "DTX_ROOT": ROOT, "DTX_OUT": str(tmp_path)})
        procs.append(subprocess.Popen([sys.executable, str(script)],
                                      env=env))
    for p in procs:
        assert p.wait(timeout=300) == 0
    out = json.load(open(tmp_path / "ad_out.json"))
    assert out["err"] < 1e-4 * max(1.0, out["ref"])


def test_streaming_matches_nonstreaming(tmp_path):
    """SSE streaming (stream=true) yields the same completion text as
    the non-streaming endpoint, chunk by chunk."""
    import json as _json
    import subprocess
    import time
    import urllib.request

    port = 18991
    env = dict(os.environ)
    env.update({"PYTHONPATH": ROOT})
    proc = subprocess.Popen(
        [sys.executable, "-m", "datatunerx_amd.serve.server",
         "--model", "llama-tiny", "--port", str(port),
         "--template", "vanilla"], env=env)
    try:
        body = _json.dumps({"messages": [{"role": "user", "content": "hi"}],
                            "max_tokens": 8}).encode()
        deadline = time.time() + 120
        resp = None
        while time.time() < deadline:
            try:
                req = urllib.request.Request(
                    f"http://127.0.0.1:{port}/chat/completions", body,
                    {"Content-Type": "application/json"})
                resp = _json.load(urllib.request.urlopen(req, timeout=10))
                break
            except Exception:
                if proc.poll() is not None:
                    raise AssertionError("server died early")
                time.sleep(1.0)
        assert resp is not None
        full = resp["choices"][0]["message"]["content"]

        sbody = _json.dumps({"messages": [{"role": "user", "content": "hi"}],
                             "max_tokens": 8, "stream": True}).encode()
        req = urllib.request.Request(
            f"http://127.0.0.1:{port}/chat/completions", sbody,
            {"Content-Type": "application/json"})
        streamed = ""
        n_chunks = 0
        with urllib.request.urlopen(req, timeout=60) as r:
            assert r.headers["Content-Type"].startswith("text/event-stream")
            for line in r:
                line = line.decode().strip()
                if not line.startswith("data: "):
                    continue
                payload = line[len("data: "):]
                if payload == "[DONE]":
                    break
                streamed += _json.loads(payload)[
                    "choices"][0]["delta"]["content"]
                n_chunks += 1
        assert streamed == full
        assert n_chunks >= 2, "should stream multiple chunks"
    finally:
        proc.kill()


HF_TP_WORKER = r"""
import os, sys
import torch
sys.path.insert(0, os.environ["DTX_ROOT"])
import torch.distributed as dist
from datatunerx_amd.models.hf_io import load_hf_config
from datatunerx_amd.parallel.tp import build_tp_llama, load_hf_weights_tp

rank = int(os.environ["RANK"]); world = int(os.environ["WORLD_SIZE"])
dist.init_process_group("gloo", rank=rank, world_size=world)
cfg = load_hf_config(os.environ["DTX_HF_DIR"])
model = build_tp_llama(cfg, rank, world, lora=False,
                       dtype=torch.float32,
                       device=torch.device("cpu"))
n = load_hf_weights_tp(model, os.environ["DTX_HF_DIR"], cfg, rank, world)
assert n > 0, n
model.eval()
torch.manual_seed(0)
ids = torch.randint(0, cfg.vocab_size, (1, 9))
with torch.no_grad():
    logits = model(ids)
if rank == 0:
    torch.save(logits, os.environ["DTX_OUT"] + "/tp_logits.pt")
dist.destroy_process_group()
"""


def test_tp_real_weights_match_single_process(tmp_path):
    """2-rank TP sharding of a REAL HF-format checkpoint produces the
    same logits as the single-process hf_io load (the 13B
    inference-compare-on-real-weights contract, gloo on CPU)."""
    import subprocess

    transformers = pytest.importorskip("transformers")
    d = str(tmp_path / "hf")
    hc = transformers.LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4,
        num_key_value_heads=2, max_position_embeddings=64,
        attn_implementation="eager")
    torch.manual_seed(5)
    transformers.LlamaForCausalLM(hc).save_pretrained(
        d, safe_serialization=True)

    script = str(tmp_path / "w.py")
    with open(script, "w") as f:
        f.write(HF_TP_WORKER)
    _port = str(free_port())
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update({"RANK": str(rank), "WORLD_SIZE": "2",
                    "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": _port,
                    "DTX_ROOT": ROOT, "DTX_OUT": str(tmp_path),
                    "DTX_HF_DIR": d})
        procs.append(subprocess.Popen([sys.executable, script], env=env))
    for p in procs:
        assert p.wait(timeout=180) == 0
    tp_logits = torch.load(str(tmp_path / "tp_logits.pt"))

    from datatunerx_amd.models import LlamaForCausalLM
    from datatunerx_amd.models.hf_io import load_hf_config, load_hf_weights
    cfg = load_hf_config(d)
    single = LlamaForCausalLM(cfg, lora=False, dtype=torch.float32)
    load_hf_weights(single, d)
    single.eval()
    torch.manual_seed(0)
    ids = torch.randint(0, cfg.vocab_size, (1, 9))
    with torch.no_grad():
        want = single(ids)
    assert torch.allclose(tp_logits, want, atol=1e-4), \
        (tp_logits - want).abs().max()
