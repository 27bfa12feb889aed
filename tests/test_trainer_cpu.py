"""Trainer plumbing on CPU: loss decreases, checkpoints have the HF
adapter layout, resume restores optimizer state (BASELINE configs[0])."""

import json
import os

import torch

from datatunerx_amd.data.dataset import SFTDataset
from datatunerx_amd.models import (GPT2Config, GPT2ForCausalLM, LlamaConfig,
                                   LlamaForCausalLM, load_adapter)
from datatunerx_amd.train.trainer import SFTTrainer, TrainerConfig, lr_at

torch.manual_seed(0)


def _tiny_trainer(tmp_path, steps=8, model=None, **kw):
    cfg = LlamaConfig.tiny()
    model = model or LlamaForCausalLM(cfg, lora=True,
                                      dtype=torch.float32).init_random()
    ds = SFTDataset.synthetic(64, 48, 512, seed=0)
    tcfg = TrainerConfig(output_dir=str(tmp_path), max_steps=steps,
                         micro_batch_size=4, logging_steps=2,
                         learning_rate=1e-3, **kw)
    return SFTTrainer(model, ds, tcfg,
                      eval_dataset=SFTDataset.synthetic(8, 48, 512, seed=9))


def test_loss_decreases(tmp_path):
    tr = _tiny_trainer(tmp_path, steps=12)
    first_losses, last_losses = [], []
    it = iter(tr.train_loader)
    for i in range(12):
        loss = tr.train_step([next(it)])
        (first_losses if i < 3 else last_losses).append(loss)
    assert sum(last_losses[-3:]) / 3 < sum(first_losses) / 3


def test_full_param_loss_decreases(tmp_path):
    """Full-parameter SFT path (reference finetuning_type=full): every
    weight trainable, overlap-bucketed optimizer auto-selected."""
    cfg = LlamaConfig.tiny()
    model = LlamaForCausalLM(cfg, lora=False, dtype=torch.float32,
                             train_base=True).init_random()
    n_train = sum(p.numel() for _, p in model.trainable_parameters())
    n_all = sum(p.numel() for p in model.parameters())
    assert n_train == n_all
    ds = SFTDataset.synthetic(64, 48, 512, seed=0)
    tr = SFTTrainer(model, ds,
                    TrainerConfig(output_dir=str(tmp_path), max_steps=20,
                                  micro_batch_size=4, logging_steps=0,
                                  lr_scheduler_type="constant",
                                  learning_rate=2e-3))
    # overfit one fixed batch: full-param must memorize it
    batch = next(iter(tr.train_loader))
    losses = [tr.train_step([batch]) for _ in range(20)]
    assert losses[-1] < losses[0] - 0.5
    # full (non-LoRA) checkpoint writes a plain safetensors state dict
    out = tr.save_checkpoint(str(tmp_path / "ck"))
    assert os.path.exists(os.path.join(out, "model.safetensors"))


def test_jsonl_and_eval_metrics(tmp_path):
    tr = _tiny_trainer(tmp_path, steps=4)
    tr.train()
    log = os.path.join(str(tmp_path), "watch", "trainer_log.jsonl")
    rows = [json.loads(l) for l in open(log)]
    assert rows and {"current_steps", "loss", "learning_rate",
                     "epoch"} <= set(rows[0])
    ev = os.path.join(str(tmp_path), "watch", "eval_log.jsonl")
    erows = [json.loads(l) for l in open(ev)]
    assert "eval_perplexity" in erows[-1]
    import math
    assert abs(erows[-1]["eval_perplexity"] -
               math.exp(erows[-1]["eval_loss"])) < 1e-2


def test_adapter_checkpoint_layout(tmp_path):
    tr = _tiny_trainer(tmp_path, steps=2)
    tr.train()
    out = tr.save_checkpoint(str(tmp_path / "ckpt"))
    assert os.path.exists(os.path.join(out, "adapter_config.json"))
    assert os.path.exists(os.path.join(out, "adapter_model.safetensors"))
    cfg = json.load(open(os.path.join(out, "adapter_config.json")))
    assert cfg["peft_type"] == "LORA"
    assert cfg["r"] == 8
    from safetensors.torch import load_file
    sd = load_file(os.path.join(out, "adapter_model.safetensors"))
    a_keys = [k for k in sd if k.endswith("lora_A.weight")]
    b_keys = [k for k in sd if k.endswith("lora_B.weight")]
    assert a_keys and len(a_keys) == len(b_keys)
    assert all(k.startswith("base_model.model.") for k in sd)
    # PEFT layouts: A [r, in], B [out, r]
    assert sd[a_keys[0]].shape[0] == 8
    assert sd[b_keys[0]].shape[1] == 8


def test_adapter_roundtrip(tmp_path):
    tr = _tiny_trainer(tmp_path, steps=2)
    tr.train()
    out = tr.save_checkpoint(str(tmp_path / "ckpt"))
    model2 = LlamaForCausalLM(LlamaConfig.tiny(), lora=True,
                              dtype=torch.float32).init_random(seed=5)
    n = load_adapter(model2, out)
    assert n > 0
    for (n1, p1), (n2, p2) in zip(
            sorted(tr.model.named_parameters()),
            sorted(model2.named_parameters())):
        if "lora" in n1:
            assert torch.allclose(p1, p2), n1


def test_resume_restores_state(tmp_path):
    tr = _tiny_trainer(tmp_path, steps=4)
    tr.train()
    out = tr.save_checkpoint(str(tmp_path / "ckpt"))
    tr2 = _tiny_trainer(tmp_path / "b", steps=4)
    tr2.load_checkpoint(out)
    assert tr2.global_step == tr.global_step
    assert torch.allclose(tr2.opt.m, tr.opt.m)
    assert torch.allclose(tr2.opt.master, tr.opt.master)


def test_lr_schedule():
    total, base = 100, 1.0
    assert lr_at(0, total, base, 0.1, "cosine") < base / 2
    assert abs(lr_at(10, total, base, 0.1, "cosine") - base) < 1e-6
    assert lr_at(99, total, base, 0.1, "cosine") < 0.01
    assert lr_at(99, total, base, 0.1, "constant") == base


def test_gpt2_plumbing(tmp_path):
    model = GPT2ForCausalLM(GPT2Config.tiny()).init_random()
    ds = SFTDataset.synthetic(16, 32, 512)
    tcfg = TrainerConfig(output_dir=str(tmp_path), max_steps=2,
                         micro_batch_size=2, logging_steps=1)
    tr = SFTTrainer(model, ds, tcfg)
    loss = tr.train()
    assert loss == loss  # finite
    out = tr.save_checkpoint(str(tmp_path / "ckpt"))
    assert os.path.exists(os.path.join(out, "adapter_model.safetensors"))


def test_grad_accumulation_equivalence(tmp_path):
    """2 micro-batches × accum ≡ 1 batch of 2× size (fp32 CPU)."""
    torch.manual_seed(3)
    cfg = LlamaConfig.tiny()
    m1 = LlamaForCausalLM(cfg, lora=True, dtype=torch.float32).init_random()
    m2 = LlamaForCausalLM(cfg, lora=True, dtype=torch.float32).init_random()
    ds = SFTDataset.synthetic(16, 32, 512)
    t1 = SFTTrainer(m1, ds, TrainerConfig(output_dir=str(tmp_path / "a"),
                                          max_steps=1, micro_batch_size=2,
                                          gradient_accumulation_steps=2,
                                          logging_steps=0))
    t2 = SFTTrainer(m2, ds, TrainerConfig(output_dir=str(tmp_path / "b"),
                                          max_steps=1, micro_batch_size=4,
                                          logging_steps=0))
    from datatunerx_amd.data.dataset import collate
    b4 = collate([ds[i] for i in range(4)])
    t1.train_step([collate([ds[0], ds[1]]), collate([ds[2], ds[3]])])
    t2.train_step([b4])
    g1 = t1.opt.master
    g2 = t2.opt.master
    assert torch.allclose(g1, g2, atol=2e-5)


def test_quantized_base_training(tmp_path):
    """int8/int4 weight-only quantization of the frozen base (reference
    bitsandbytes path, Hyperparameter int4/int8 flags): quantized model
    stays close to the bf16 one and LoRA still trains."""
    from datatunerx_amd.models.quant import quantize_model_
    cfg = LlamaConfig.tiny()
    model = LlamaForCausalLM(cfg, lora=True,
                             dtype=torch.float32).init_random()
    ids = torch.randint(0, cfg.vocab_size, (1, 32))
    with torch.no_grad():
        ref_logits = model(ids)
    n = quantize_model_(model, bits=8)
    assert n > 0
    with torch.no_grad():
        q_logits = model(ids)
    err = (q_logits - ref_logits).abs().max() / ref_logits.abs().max()
    assert err < 0.2, f"int8 rel err {err}"
    ds = SFTDataset.synthetic(16, 32, cfg.vocab_size, seed=0)
    tr = SFTTrainer(model, ds,
                    TrainerConfig(output_dir=str(tmp_path), max_steps=2,
                                  micro_batch_size=2, logging_steps=0))
    it = iter(tr.train_loader)
    loss = tr.train_step([next(it)])
    assert loss == loss
    # int4 path roundtrips too
    m4 = LlamaForCausalLM(cfg, lora=False,
                          dtype=torch.float32).init_random()
    quantize_model_(m4, bits=4)
    with torch.no_grad():
        l4 = m4(ids)
    assert torch.isfinite(l4).all()


def test_predict_writes_jsonl(tmp_path):
    """predict() greedy-generates from the -100-masked prompt prefix and
    writes generated_predictions.jsonl (trainer.py:405-489 parity)."""
    tr = _tiny_trainer(tmp_path, steps=1)
    tr.train()
    ds = SFTDataset.synthetic(3, 24, 512, seed=1)
    res = tr.predict(ds, max_new_tokens=4, eos_token_id=None)
    assert len(res) == 3
    assert all(len(r["predict_ids"]) == 4 for r in res)
    # with eos stopping enabled, output never contains eos and is <= 4
    res2 = tr.predict(ds, max_new_tokens=4)
    assert all(2 not in r["predict_ids"] and len(r["predict_ids"]) <= 4
               for r in res2)
    # eos-stopped output is a prefix of the unstopped greedy output
    # (same KV-cache decode, so tokens must agree until the stop)
    for r, q in zip(res, res2):
        assert r["predict_ids"][:len(q["predict_ids"])] == q["predict_ids"]
    out = os.path.join(str(tmp_path), "generated_predictions.jsonl")
    rows = [json.loads(l) for l in open(out)]
    assert len(rows) == 3 and "prompt_ids" in rows[0]


def test_generation_metrics():
    from datatunerx_amd.train.gen_metrics import (bleu, generation_metrics,
                                                  rouge_l)
    assert rouge_l([1, 2, 3], [1, 2, 3]) == 1.0
    assert rouge_l([1, 2, 3], [4, 5, 6]) == 0.0
    assert bleu([1, 2, 3, 4], [1, 2, 3, 4]) > 0.99
    assert bleu([9, 9, 9, 9], [1, 2, 3, 4]) < 0.01
    m = generation_metrics([[1, 2, 3]], [[1, 2, 4]])
    assert 0 < m["predict_rouge-l"] < 1


def test_rng_dropout_cpu_consistency():
    """The CPU splitmix64 dropout twin: deterministic per seed, correct
    keep-rate, and the seed-mode reference ops equal the explicit-mask
    reference ops (the invariant the fused GPU kernels also satisfy)."""
    import torch

    from datatunerx_amd.ops import reference as ref
    M, K, r, seed, keep = 128, 256, 8, 77, 0.8
    m1 = ref.dropout_mask(M, K, seed, keep)
    m2 = ref.dropout_mask(M, K, seed, keep)
    assert torch.equal(m1, m2)
    assert abs((m1 > 0).float().mean().item() - keep) < 0.05
    x = torch.randn(M, K, dtype=torch.bfloat16)
    a = torch.randn(r, K, dtype=torch.bfloat16)
    t_seed = ref.lora_contract(x, a, seed=seed, keep=keep)
    t_mask = ref.lora_contract(x, a, mask=m1.to(x.dtype))
    assert torch.equal(t_seed, t_mask)


def test_residual_epilogue_fold_matches_separate_add():
    """o_proj/down_proj fold the residual into the GEMM epilogue
    (addmm); the result must equal the separate linear + add (same op
    order in fp32 on CPU)."""
    import torch

    from datatunerx_amd.models.lora import FrozenLinear
    torch.manual_seed(0)
    lin = FrozenLinear(32, 48, dtype=torch.float32)
    with torch.no_grad():
        lin.weight.copy_(torch.randn(48, 32))
    x = torch.randn(2, 5, 32)
    res = torch.randn(2, 5, 48)
    fused = lin(x, residual=res)
    sep = res + lin(x)
    assert torch.allclose(fused, sep, atol=1e-5)
    # gradient flows through both x and residual
    x2 = x.clone().requires_grad_(True)
    r2 = res.clone().requires_grad_(True)
    lin(x2, residual=r2).sum().backward()
    assert torch.allclose(r2.grad, torch.ones_like(r2))
    assert x2.grad is not None


def test_engine_stream_matches_chat_cpu():
    """chat_stream deltas concatenate to exactly chat()'s text (eager
    CPU path; the GPU graph path is covered in test_ops_gpu)."""
    import torch

    from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM
    from datatunerx_amd.serve.engine import InferenceEngine
    model = LlamaForCausalLM(LlamaConfig.tiny(), lora=False,
                             dtype=torch.float32).init_random(seed=0)
    model.eval()
    eng = InferenceEngine(model, template="vanilla",
                          device=torch.device("cpu"))
    msgs = [{"role": "user", "content": "hello"}]
    full = eng.chat(msgs, 8)
    streamed = "".join(eng.chat_stream(msgs, 8))
    assert streamed == full


def test_model_loss_fused_ce_env(monkeypatch):
    """DTX_FUSED_CE=1 routes the model loss through the chunked fused
    path and matches the materialized default."""
    import torch

    from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM
    torch.manual_seed(0)
    m = LlamaForCausalLM(LlamaConfig.tiny(), lora=True,
                         dtype=torch.float32).init_random()
    ids = torch.randint(3, 512, (2, 32))
    base = float(m(ids, labels=ids.clone()).detach())
    monkeypatch.setenv("DTX_FUSED_CE", "1")
    fused = float(m(ids, labels=ids.clone()).detach())
    assert abs(base - fused) < 1e-4


def test_fused_qkv_gateup_grads_match_unfused():
    """The QKVProj / PairedFrozenGemm nodes produce the same loss and
    LoRA gradients as the per-module path (dropout off; the fused
    backward accumulates dgrads via addmm_ instead of autograd adds)."""
    import torch

    from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM
    torch.manual_seed(0)
    cfg = LlamaConfig.tiny(lora_dropout=0.0)
    m = LlamaForCausalLM(cfg, lora=True, dtype=torch.float32).init_random()
    ids = torch.randint(3, cfg.vocab_size, (2, 48))
    for layer in m.layers:
        assert layer.self_attn._fused_qkv()
    loss = m(ids, labels=ids.clone())
    loss.backward()
    g_fused = {n: p.grad.clone() for n, p in m.trainable_parameters()}
    for _, p in m.trainable_parameters():
        p.grad = None

    # disable the fused nodes by marking k_proj's weight as "trainable"
    # in the gate only (requires_grad flips the _fused_qkv condition);
    # restore after the forward so the optimizer view stays identical
    for layer in m.layers:
        layer.self_attn._fused_qkv = lambda: False
        layer.mlp.gate_proj.weight.requires_grad_(True)
    loss2 = m(ids, labels=ids.clone())
    loss2.backward()
    assert torch.allclose(loss, loss2, atol=1e-6)
    for n, p in m.trainable_parameters():
        if p.grad is None or n not in g_fused:
            continue                  # gate_proj became "trainable" only
            # to flip the fused gate; it has no fused-side grad to compare
        assert torch.allclose(g_fused[n], p.grad, atol=1e-5), n


def test_run_do_predict_cli(tmp_path):
    """--do_predict true writes generated_predictions.jsonl and reports
    rouge/bleu in the status file (GenEvalSeq2SeqTrainer flow)."""
    import csv

    from datatunerx_amd.train.run import main as train_main
    p = tmp_path / "d.csv"
    with open(p, "w", newline="") as f:
        w = csv.writer(f)
        w.writerow(["instruction", "response"])
        for i in range(8):
            w.writerow([f"say number {i}", f"number {i}"])
    out = str(tmp_path / "out")
    status = str(tmp_path / "st.json")
    os.environ["DTX_STATUS_FILE"] = status
    try:
        rc = train_main([
            "--model_name_or_path", "llama-tiny", "--output_dir", out,
            "--dataset_path", str(p), "--eval_dataset_path", str(p),
            "--max_steps", "2", "--per_device_train_batch_size", "2",
            "--block_size", "64", "--logging_steps", "0",
            "--do_predict", "true", "--predict_max_new_tokens", "4"])
    finally:
        del os.environ["DTX_STATUS_FILE"]
    assert rc == 0
    st = json.load(open(status))
    assert st["state"] == "Successful"
    assert "predict_rouge-l" in st["predict"]
    gen = os.path.join(out, "generated_predictions.jsonl")
    rows = [json.loads(l) for l in open(gen)]
    assert len(rows) == 8 and "predict_ids" in rows[0]


def test_ignored_flags_warn(tmp_path, capsys):
    from datatunerx_amd.train.run import main as train_main
    rc = train_main([
        "--model_name_or_path", "llama-tiny", "--output_dir",
        str(tmp_path / "o"), "--max_steps", "1",
        "--per_device_train_batch_size", "2", "--synthetic_examples",
        "8", "--block_size", "32", "--logging_steps", "0",
        "--optim", "sgd", "--fp16", "true", "--shift_attn", "true"])
    assert rc == 0
    out = capsys.readouterr().out
    assert "--optim 'sgd' ignored" in out
    assert "--fp16 ignored" in out
    assert "--shift_attn ignored" in out


def test_invalid_optimizer_mode_rejected(tmp_path):
    import pytest
    with pytest.raises(ValueError, match="optimizer_mode"):
        _tiny_trainer(tmp_path, steps=1, optimizer_mode="sgdish")
