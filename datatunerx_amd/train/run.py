"""Trainer process entrypoint — the replacement for the reference's
RayJob entrypoint `python /tuning/train.py ...` (finetune_controller.go:
451-516 builds the flag list; train.py:308-390 is the driver).

Launched by the Finetune controller as one process per GPU (torchrun-
style env: RANK/WORLD_SIZE/MASTER_ADDR/MASTER_PORT, HIP_VISIBLE_DEVICES
pinned per rank). Writes a status JSON file the controller polls —
replacing the reference's pod-exec of /home/ray/checkpoint_path
(finetune_controller.go:278-305).
"""

from __future__ import annotations

import json
import os
import sys
import traceback


def write_status(path: str, state: str, **kw):
    if not path:
        return
    tmp = path + ".tmp"
    with open(tmp, "w") as f:
        json.dump({"state": state, **kw}, f)
    os.replace(tmp, path)


def main(argv=None):
    import torch

    from ..data.dataset import ByteTokenizer, SFTDataset
    from ..models import (GPT2Config, GPT2ForCausalLM, LlamaConfig,
                          LlamaForCausalLM)
    from ..parallel.ddp import init_distributed, is_main
    from .args import get_train_args
    from .trainer import SFTTrainer, TrainerConfig

    margs, fargs, dargs = get_train_args(argv)
    status_file = os.environ.get("DTX_STATUS_FILE", "")
    rank, world, local_rank, device = init_distributed()
    if rank == 0:
        # declared-for-CR-parity flags this trainer does not act on:
        # say so loudly rather than silently ignoring a request
        if fargs.optim not in ("adamw_torch", "adamw"):
            print(f"warning: --optim {fargs.optim!r} ignored — the "
                  f"fused AdamW kernel is the only optimizer")
        if fargs.fp16:
            print("warning: --fp16 ignored — compute dtype is bf16 "
                  "(MI355X-native; >= the reference's fp16)")
        if margs.rope_scaling:
            print("warning: --rope_scaling ignored (dormant in the "
                  "reference too, parser.py:60-68)")
        if margs.shift_attn:
            print("warning: --shift_attn ignored (dormant in the "
                  "reference too, parser.py:70-73)")
    try:
        if rank == 0:
            write_status(status_file, "Running")
        torch.manual_seed(fargs.seed)
        dtype = torch.bfloat16 if device.type == "cuda" else torch.float32
        name = margs.model_name_or_path
        lora = fargs.finetuning_type == "lora"
        full = fargs.finetuning_type == "full"
        lora_kw = dict(lora_r=fargs.lora_rank, lora_alpha=fargs.lora_alpha,
                       lora_dropout=fargs.lora_dropout,
                       lora_targets=fargs.lora_targets)
        from ..models.hf_io import (is_hf_model_dir, load_hf_config,
                                    load_hf_weights, load_tokenizer)
        hf_dir = is_hf_model_dir(name)
        with torch.device(device):
            if hf_dir:
                # real HF-format checkpoint from a local dir (the
                # reference's from_pretrained contract, train.py:236-242)
                model = LlamaForCausalLM(load_hf_config(
                    name,
                    gradient_checkpointing=fargs.gradient_checkpointing,
                    **lora_kw), lora=lora, dtype=dtype, train_base=full)
            elif name in ("llama2-7b", "llama-2-7b"):
                model = LlamaForCausalLM(LlamaConfig.llama2_7b(
                    gradient_checkpointing=fargs.gradient_checkpointing,
                    **lora_kw), lora=lora, dtype=dtype, train_base=full)
            elif name in ("llama2-13b", "llama-2-13b"):
                model = LlamaForCausalLM(LlamaConfig.llama2_13b(
                    gradient_checkpointing=fargs.gradient_checkpointing,
                    **lora_kw), lora=lora, dtype=dtype, train_base=full)
            elif name in ("llama3-8b", "llama-3-8b"):
                model = LlamaForCausalLM(LlamaConfig.llama3_8b(
                    gradient_checkpointing=fargs.gradient_checkpointing,
                    **lora_kw), lora=lora, dtype=dtype, train_base=full)
            elif name == "llama-tiny":
                model = LlamaForCausalLM(LlamaConfig.tiny(**lora_kw),
                                         lora=lora, dtype=dtype,
                                         train_base=full)
            elif name == "llama-mini":
                model = LlamaForCausalLM(LlamaConfig.mini(**lora_kw),
                                         lora=lora, dtype=dtype,
                                         train_base=full)
            elif name in ("gpt2-small", "gpt2"):
                model = GPT2ForCausalLM(GPT2Config.small(
                    lora_r=fargs.lora_rank, lora_alpha=fargs.lora_alpha,
                    lora_dropout=fargs.lora_dropout), dtype=dtype)
            elif name == "gpt2-tiny":
                model = GPT2ForCausalLM(GPT2Config.tiny(), dtype=dtype)
            else:
                raise SystemExit(f"unknown model {name!r}")
        # no network: checkpoints load from local dirs, else random init
        if hf_dir:
            n_t = load_hf_weights(model, name)
            if rank == 0:
                print(f"loaded {n_t} weight tensors from {name}")
        elif margs.checkpoint_dir and os.path.isdir(margs.checkpoint_dir):
            from safetensors.torch import load_file
            sd = load_file(os.path.join(margs.checkpoint_dir,
                                        "model.safetensors"))
            model.load_state_dict(sd, strict=False)
        else:
            model.init_random(seed=fargs.seed)
        if margs.quantization in ("int4", "int8"):
            from ..models.quant import quantize_model_
            n_q = quantize_model_(model,
                                  bits=4 if margs.quantization == "int4"
                                  else 8)
            if rank == 0:
                print(f"quantized {n_q} frozen base layers to "
                      f"{margs.quantization}")

        tok = load_tokenizer(name)
        vocab = getattr(model.cfg, "vocab_size")
        if fargs.stage not in ("sft", "pt", "dpo"):
            raise SystemExit(
                f"stage {fargs.stage!r} not supported (sft, pt, dpo)")
        pt = fargs.stage == "pt"
        dpo = fargs.stage == "dpo"
        if dpo:
            from ..data.preference import PreferenceDataset
            if dargs.dataset_path and os.path.exists(dargs.dataset_path):
                ds = PreferenceDataset.from_csv(
                    dargs.dataset_path, tok,
                    column_map={"instruction": dargs.instruction_column,
                                "chosen": dargs.chosen_column,
                                "rejected": dargs.rejected_column},
                    template_name=dargs.prompt_template,
                    cutoff_len=dargs.block_size)
            else:
                n = dargs.synthetic_examples or 256
                ds = PreferenceDataset.synthetic(n, dargs.block_size,
                                                 vocab, seed=fargs.seed)
        elif dargs.dataset_path and os.path.exists(dargs.dataset_path):
            from ..data.dataset import read_csv_rows
            cmap = {"instruction": dargs.instruction_column,
                    "response": dargs.response_column}
            if pt:
                # pretraining: pack text blocks, train every position
                ds = SFTDataset.from_rows_pt(
                    read_csv_rows(dargs.dataset_path, cmap), tok,
                    cutoff_len=dargs.block_size)
            else:
                ds = SFTDataset.from_csv(
                    dargs.dataset_path, tok, column_map=cmap,
                    template_name=dargs.prompt_template,
                    cutoff_len=dargs.block_size)
        else:
            n = dargs.synthetic_examples or 256
            ds = SFTDataset.synthetic(n, dargs.block_size, vocab,
                                      seed=fargs.seed,
                                      mask_frac=0.0 if pt else 0.25)
        eval_ds = None
        if dpo and dargs.eval_dataset_path and \
                os.path.exists(dargs.eval_dataset_path):
            from ..data.preference import PreferenceDataset
            eval_ds = PreferenceDataset.from_csv(
                dargs.eval_dataset_path, tok,
                column_map={"instruction": dargs.instruction_column,
                            "chosen": dargs.chosen_column,
                            "rejected": dargs.rejected_column},
                template_name=dargs.prompt_template,
                cutoff_len=dargs.block_size)
        elif dargs.eval_dataset_path and \
                os.path.exists(dargs.eval_dataset_path):
            from ..data.dataset import read_csv_rows
            cmap = {"instruction": dargs.instruction_column,
                    "response": dargs.response_column}
            if pt:               # eval perplexity on packed blocks too
                eval_ds = SFTDataset.from_rows_pt(
                    read_csv_rows(dargs.eval_dataset_path, cmap), tok,
                    cutoff_len=dargs.block_size)
            else:
                eval_ds = SFTDataset.from_csv(
                    dargs.eval_dataset_path, tok, column_map=cmap,
                    template_name=dargs.prompt_template,
                    cutoff_len=dargs.block_size)

        tcfg = TrainerConfig(
            output_dir=fargs.output_dir,
            learning_rate=fargs.learning_rate,
            weight_decay=fargs.weight_decay,
            max_grad_norm=fargs.max_grad_norm,
            warmup_ratio=fargs.warmup_ratio,
            lr_scheduler_type=fargs.lr_scheduler_type,
            num_train_epochs=fargs.num_train_epochs,
            max_steps=fargs.max_steps,
            micro_batch_size=fargs.per_device_train_batch_size,
            eval_batch_size=fargs.per_device_eval_batch_size,
            gradient_accumulation_steps=fargs.gradient_accumulation_steps,
            logging_steps=fargs.logging_steps,
            eval_steps=fargs.eval_steps, save_steps=fargs.save_steps,
            seed=fargs.seed,
            metrics_export_address=fargs.metrics_export_address,
            uid=fargs.uid, lora_r=fargs.lora_rank,
            lora_alpha=fargs.lora_alpha, lora_dropout=fargs.lora_dropout,
            lora_targets=fargs.lora_targets, base_model=name)
        if dpo:
            from .trainer import DPOTrainer
            trainer = DPOTrainer(model, ds, tcfg, eval_dataset=eval_ds,
                                 device=device, rank=rank,
                                 world_size=world,
                                 pad_token_id=tok.pad_token_id,
                                 beta=fargs.dpo_beta)
        else:
            trainer = SFTTrainer(model, ds, tcfg, eval_dataset=eval_ds,
                                 device=device, rank=rank,
                                 world_size=world,
                                 pad_token_id=tok.pad_token_id)
        final_loss = trainer.train()
        ckpt_path = None
        if is_main():
            ckpt_path = trainer.save_checkpoint(
                os.path.join(fargs.output_dir, "checkpoint"))
            if fargs.storage_path:
                # "upload": copy to the storage root (replaces S3 push,
                # train.py:302-305)
                import shutil
                dst = os.path.join(fargs.storage_path, fargs.uid or "ckpt",
                                   "checkpoint")
                if os.path.abspath(dst) != os.path.abspath(ckpt_path):
                    shutil.copytree(ckpt_path, dst, dirs_exist_ok=True)
                    ckpt_path = dst
            predict_metrics = None
            if fargs.do_predict and eval_ds is not None and not dpo:
                # generation eval: generated_predictions.jsonl + token
                # ROUGE-L/BLEU (GenEvalSeq2SeqTrainer parity)
                res = trainer.predict(
                    eval_ds, max_new_tokens=fargs.predict_max_new_tokens,
                    eos_token_id=tok.eos_token_id)
                from .gen_metrics import generation_metrics
                predict_metrics = generation_metrics(
                    [r["predict_ids"] for r in res],
                    [r["label_ids"] for r in res])
            write_status(status_file, "Successful",
                         checkpoint_path=ckpt_path,
                         final_loss=final_loss,
                         eval=trainer.evaluate() if eval_ds else None,
                         predict=predict_metrics)
        import torch.distributed as dist
        if dist.is_initialized():
            dist.barrier()
            dist.destroy_process_group()
        return 0
    except Exception:
        traceback.print_exc()
        if rank == 0:
            write_status(status_file, "Failed", error=traceback.format_exc())
        return 1


if __name__ == "__main__":
    sys.exit(main())
