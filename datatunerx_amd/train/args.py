"""Trainer argument dataclasses (parity with cmd/tuning/parser.py:12-266;
the operator's CR-field→flag mapping is finetune_controller.go:451-516)."""

from __future__ import annotations

import argparse
from dataclasses import dataclass, field, fields
from typing import Optional


@dataclass
class ModelArguments:
    model_name_or_path: str = "llama2-7b"
    quantization: Optional[str] = None          # int4 / int8 (phase-2)
    rope_scaling: Optional[str] = None          # dormant in reference too
    flash_attn: bool = True                     # our FA kernels are default
    shift_attn: bool = False                    # dormant (parser.py:70-73)
    checkpoint_dir: Optional[str] = None


@dataclass
class FinetuningArguments:
    stage: str = "sft"        # sft (prompt-masked) | pt (packed blocks,
                              # all positions trained) | dpo (preference
                              # pairs, adapters-off reference policy);
                              # rm/ppo are rejected with a clear error
                              # (the reference declares but never runs
                              # any of these — parser.py:112-221)
    dpo_beta: float = 0.1
    finetuning_type: str = "lora"               # lora / full
    lora_rank: int = 8
    lora_alpha: float = 32.0
    lora_dropout: float = 0.1
    lora_target: str = "q_proj,v_proj"          # finetune_controller.go:482
    num_workers: int = 1
    storage_path: Optional[str] = None
    metrics_export_address: Optional[str] = None
    uid: str = ""
    output_dir: str = "./output"
    learning_rate: float = 2e-4
    num_train_epochs: float = 1.0
    max_steps: int = -1
    per_device_train_batch_size: int = 4
    per_device_eval_batch_size: int = 4
    gradient_accumulation_steps: int = 1
    warmup_ratio: float = 0.03
    weight_decay: float = 0.0
    lr_scheduler_type: str = "cosine"
    optim: str = "adamw_torch"
    logging_steps: int = 10
    save_steps: int = 0                         # 0 = terminal save only
    max_grad_norm: float = 1.0
    seed: int = 42
    bf16: bool = True
    fp16: bool = False
    gradient_checkpointing: bool = False
    eval_steps: int = 0
    do_predict: bool = False      # after training, greedy-generate from
                                  # the eval set's prompts and write
                                  # generated_predictions.jsonl + token
                                  # ROUGE-L/BLEU (the reference's
                                  # GenEvalSeq2SeqTrainer flow,
                                  # cmd/tuning/trainer.py:29-172)
    predict_max_new_tokens: int = 64

    @property
    def lora_targets(self):
        return tuple(t.strip() for t in self.lora_target.split(",") if t.strip())


@dataclass
class DataArguments:
    dataset_path: Optional[str] = None
    eval_dataset_path: Optional[str] = None
    instruction_column: str = "instruction"
    response_column: str = "response"
    chosen_column: str = "chosen"               # stage=dpo
    rejected_column: str = "rejected"           # stage=dpo
    prompt_template: str = "llama2"
    block_size: int = 1024                      # cutoff_len (train.py:49-51)
    synthetic_examples: int = 0                 # >0: synthetic data


def _add_dataclass_args(parser: argparse.ArgumentParser, dc) -> None:
    for f in fields(dc):
        name = "--" + f.name
        if f.type in ("bool", bool) or isinstance(f.default, bool):
            parser.add_argument(name, type=lambda s: s.lower() in
                                ("1", "true", "yes"), default=f.default)
        else:
            typ = str
            if isinstance(f.default, int):
                typ = int
            elif isinstance(f.default, float):
                typ = float
            parser.add_argument(name, type=typ, default=f.default)


def get_train_args(argv=None):
    """parser.py:250-266 equivalent."""
    p = argparse.ArgumentParser("datatunerx_amd trainer")
    for dc in (ModelArguments, FinetuningArguments, DataArguments):
        _add_dataclass_args(p, dc)
    ns, _ = p.parse_known_args(argv)
    def pick(dc):
        names = {f.name for f in fields(dc)}
        return dc(**{k: v for k, v in vars(ns).items() if k in names})
    return pick(ModelArguments), pick(FinetuningArguments), pick(DataArguments)
