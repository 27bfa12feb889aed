"""DPO (stage=dpo) tests: adapters-off reference-policy exactness,
loss-vs-pure-torch equivalence, learning on synthetic preferences, CLI."""

import json
import os

import pytest
import torch
import torch.nn.functional as F

from datatunerx_amd.data.preference import (PreferenceDataset,
                                            collate_preference)
from datatunerx_amd.models import LlamaConfig, LlamaForCausalLM
from datatunerx_amd.train.trainer import (DPOTrainer, SFTTrainer,
                                          TrainerConfig, _AdaptersDisabled)


def _models(seed=0):
    cfg = LlamaConfig.tiny(lora_dropout=0.0)
    torch.manual_seed(seed)
    lora = LlamaForCausalLM(cfg, lora=True, dtype=torch.float32)
    lora.init_random(seed=3 + seed)
    base = LlamaForCausalLM(cfg, lora=False, dtype=torch.float32)
    # same BASE weights by name (same-seed init_random would diverge:
    # the lora model's extra params consume generator draws)
    base.load_state_dict({k: v for k, v in lora.state_dict().items()
                          if "lora_" not in k and "_wt" not in k},
                         strict=False)
    return lora, base


def test_adapters_disabled_equals_base_model():
    """scale=0 makes the LoRA model compute EXACTLY the frozen base —
    the DPO reference policy without a second weight copy."""
    lora, base = _models()
    # make adapters matter when enabled (B starts at zero)
    with torch.no_grad():
        for n, p in lora.named_parameters():
            if "lora_B" in n:
                p.normal_(0, 0.1)
    ids = torch.randint(3, 500, (2, 33))
    labels = ids.clone()
    labels[:, :9] = -100
    with torch.no_grad():
        on = lora.sequence_logprobs(ids, labels)
        with _AdaptersDisabled(lora):
            off = lora.sequence_logprobs(ids, labels)
        ref = base.sequence_logprobs(ids, labels)
    assert not torch.allclose(on, off)          # adapters were active
    assert torch.allclose(off, ref, atol=1e-5), (off - ref).abs().max()
    # and the context restored the scales
    with torch.no_grad():
        assert torch.allclose(lora.sequence_logprobs(ids, labels), on)


def test_sequence_logprobs_matches_torch():
    lora, _ = _models()
    ids = torch.randint(3, 500, (2, 21))
    labels = ids.clone()
    labels[:, :5] = -100
    with torch.no_grad():
        got = lora.sequence_logprobs(ids, labels)
        logits = lora(ids)                       # [B, S, V]
        lp = F.log_softmax(logits[:, :-1].float(), dim=-1)
        t = labels[:, 1:]
        mask = t != -100
        want = (lp.gather(-1, t.clamp(min=0).unsqueeze(-1)).squeeze(-1)
                * mask).sum(1)
    assert torch.allclose(got, want, atol=1e-4), (got - want).abs().max()


def test_dpo_micro_loss_matches_pure_torch(tmp_path):
    lora, _ = _models()
    ds = PreferenceDataset.synthetic(8, 32, 500, seed=5)
    tr = DPOTrainer(lora, ds, TrainerConfig(
        output_dir=str(tmp_path), max_steps=1, micro_batch_size=4,
        logging_steps=0), beta=0.25)
    mb = collate_preference([ds[i] for i in range(4)])
    loss = tr._micro_loss(mb)
    # independent computation
    with torch.no_grad():
        def lp(model, ids, labels):
            logits = model(ids)
            l = F.log_softmax(logits[:, :-1].float(), -1)
            t = labels[:, 1:]
            m = t != -100
            return (l.gather(-1, t.clamp(min=0).unsqueeze(-1)).squeeze(-1)
                    * m).sum(1)
        pc = lp(lora, mb["chosen_ids"], mb["chosen_labels"])
        pr = lp(lora, mb["rejected_ids"], mb["rejected_labels"])
        with _AdaptersDisabled(lora):
            rc = lp(lora, mb["chosen_ids"], mb["chosen_labels"])
            rr = lp(lora, mb["rejected_ids"], mb["rejected_labels"])
        want = -F.logsigmoid(0.25 * ((pc - rc) - (pr - rr))).mean()
    assert torch.allclose(loss.detach(), want, atol=1e-4), (loss, want)
    # fresh adapters: reference == policy, so loss == -logsigmoid(0)
    assert abs(float(loss.detach()) - 0.6931) < 1e-3


def test_dpo_training_learns_preferences(tmp_path):
    lora, _ = _models(seed=1)
    ds = PreferenceDataset.synthetic(32, 32, 500, seed=7)
    tr = DPOTrainer(lora, ds, TrainerConfig(
        output_dir=str(tmp_path), max_steps=30, micro_batch_size=8,
        logging_steps=0, learning_rate=5e-3, warmup_ratio=0.0),
        beta=0.5, eval_dataset=ds)
    tr.train()
    m = tr.evaluate()
    assert m["eval_pref_accuracy"] > 0.9, m
    assert m["eval_margin"] > 0, m
    # only adapters moved (base stays the reference policy)
    assert tr.last_train_loss < 0.69


def test_dpo_requires_lora():
    _, base = _models()
    with pytest.raises(ValueError, match="LoRA"):
        DPOTrainer(base, PreferenceDataset.synthetic(4, 16, 500),
                   TrainerConfig(output_dir="/tmp/x", max_steps=1))


def test_preference_csv_and_collate(tmp_path):
    import csv
    p = tmp_path / "pref.csv"
    with open(p, "w", newline="") as f:
        w = csv.writer(f)
        w.writerow(["instruction", "chosen", "rejected"])
        for i in range(6):
            w.writerow([f"question {i}", f"good answer {i}",
                        f"bad {i}"])
    from datatunerx_amd.data.dataset import ByteTokenizer
    ds = PreferenceDataset.from_csv(str(p), ByteTokenizer(),
                                    template_name="default")
    assert len(ds) == 6
    ex = ds[0]
    assert any(l != -100 for l in ex["chosen_labels"])
    assert ex["chosen_ids"][:5] == ex["rejected_ids"][:5]  # same prompt
    b = collate_preference([ds[0], ds[1]])
    assert b["chosen_ids"].shape[0] == 2
    assert b["chosen_labels"].shape == b["chosen_ids"].shape


def test_run_stage_dpo_cli(tmp_path):
    from datatunerx_amd.train.run import main as train_main
    out = str(tmp_path / "out")
    rc = train_main([
        "--model_name_or_path", "llama-tiny", "--stage", "dpo",
        "--output_dir", out, "--max_steps", "3",
        "--per_device_train_batch_size", "4", "--synthetic_examples",
        "16", "--block_size", "64", "--logging_steps", "1",
        "--dpo_beta", "0.2", "--lora_dropout", "0.0"])
    assert rc == 0
    log = os.path.join(out, "watch", "trainer_log.jsonl")
    rows = [json.loads(l) for l in open(log)]
    assert rows[-1]["current_steps"] == 3
    assert os.path.exists(os.path.join(out, "checkpoint",
                                       "adapter_model.safetensors"))


def test_dpo_two_rank_gloo(tmp_path):
    """2-rank data-parallel DPO through train.run (gloo): both ranks
    finish, losses logged, one adapter checkpoint written."""
    import subprocess
    import sys

    from conftest import free_port
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = str(tmp_path / "out")
    port = str(free_port())
    procs = []
    for rank in range(2):
        env = dict(os.environ)
        env.update({"RANK": str(rank), "WORLD_SIZE": "2",
                    "LOCAL_RANK": str(rank), "MASTER_ADDR": "127.0.0.1",
                    "MASTER_PORT": port, "PYTHONPATH": root})
        procs.append(subprocess.Popen(
            [sys.executable, "-m", "datatunerx_amd.train.run",
             "--model_name_or_path", "llama-tiny", "--stage", "dpo",
             "--output_dir", out, "--max_steps", "3",
             "--per_device_train_batch_size", "2",
             "--synthetic_examples", "16", "--block_size", "48",
             "--logging_steps", "1", "--lora_dropout", "0.0"],
            env=env))
    for p in procs:
        assert p.wait(timeout=300) == 0
    rows = [json.loads(l) for l in
            open(os.path.join(out, "watch", "trainer_log.jsonl"))]
    assert rows[-1]["current_steps"] == 3
    assert "reward_margin" in rows[-1]
    assert os.path.exists(os.path.join(out, "checkpoint",
                                       "adapter_model.safetensors"))


def test_dpo_gpt2(tmp_path):
    """DPO on the GPT-2 family (sequence_logprobs parity)."""
    from datatunerx_amd.models import GPT2Config, GPT2ForCausalLM
    torch.manual_seed(0)
    m = GPT2ForCausalLM(GPT2Config.tiny(), dtype=torch.float32)
    m.init_random(seed=1)
    ds = PreferenceDataset.synthetic(8, 24, 250, seed=2)
    tr = DPOTrainer(m, ds, TrainerConfig(
        output_dir=str(tmp_path), max_steps=2, micro_batch_size=4,
        logging_steps=0), beta=0.3)
    it = iter(tr.train_loader)
    l0 = tr.train_step([next(it)])
    assert abs(l0 - 0.693) < 0.01          # fresh adapters: margin 0
    # independent torch check of gpt2 sequence_logprobs
    ids = torch.randint(3, 250, (2, 17))
    labels = ids.clone()
    labels[:, :4] = -100
    with torch.no_grad():
        got = m.sequence_logprobs(ids, labels)
        lp = F.log_softmax(m(ids)[:, :-1].float(), -1)
        t = labels[:, 1:]
        msk = t != -100
        want = (lp.gather(-1, t.clamp(min=0).unsqueeze(-1)).squeeze(-1)
                * msk).sum(1)
    assert torch.allclose(got, want, atol=1e-4)


def test_dpo_disables_dropout():
    """DPOTrainer forces adapter dropout off (policy/reference logprob
    comparison must be deterministic — TRL-style disable_dropout)."""
    from datatunerx_amd.models.lora import LoRALinearModule
    cfg = LlamaConfig.tiny(lora_dropout=0.3)
    m = LlamaForCausalLM(cfg, lora=True, dtype=torch.float32)
    m.init_random(seed=1)
    DPOTrainer(m, PreferenceDataset.synthetic(4, 16, 500),
               TrainerConfig(output_dir="/tmp/dpodrop", max_steps=1,
                             micro_batch_size=2, logging_steps=0))
    assert all(mod.dropout == 0.0 for mod in m.modules()
               if isinstance(mod, LoRALinearModule))
