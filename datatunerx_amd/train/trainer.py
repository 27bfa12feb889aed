"""Native SFT trainer: the replacement for the reference's Ray-wrapped
HF `SFTTrainer` (cmd/tuning/trainer.py:175-507, train.py:138-305).

Hot loop design (MI355X-first):
- bf16 compute through the fused HIP ops; frozen-base GEMMs via hipBLASLt.
- trainable params tracked in ONE flat fp32 master + m/v + grad buffer;
  gradient sync is a single fused RCCL all-reduce on the flat fp32 grad
  (LoRA is latency-bound over xGMI); fused AdamW HIP kernel updates
  master+param in one pass.
- eval adds perplexity = exp(eval_loss) (trainer.py:323-327 parity).
- metrics: jsonl under output_dir/watch/ + Prometheus remote-write
  (callback.py:95-155, prometheus/metrics.py wire contract).
- checkpoints: HF-adapter layout for LoRA (PEFT-loadable); periodic
  save + resume state (an improvement over the reference's single
  terminal save — SURVEY.md §5 Checkpoint/resume).
"""

from __future__ import annotations

import json
import math
import os
import time
from dataclasses import dataclass
from typing import Optional

import torch

from .. import ops
from ..data.dataset import ShardedLoader
from ..models.lora import LoRALinearModule, save_adapter
from ..parallel.ddp import GradSynchronizer, is_main, sync_scalar_mean


@dataclass
class TrainerConfig:
    output_dir: str = "./output"
    learning_rate: float = 2e-4
    betas: tuple = (0.9, 0.999)
    eps: float = 1e-8
    weight_decay: float = 0.0
    max_grad_norm: float = 1.0
    warmup_ratio: float = 0.03
    lr_scheduler_type: str = "cosine"      # cosine | linear | constant
    num_train_epochs: float = 1.0
    max_steps: int = -1                    # overrides epochs when > 0
    micro_batch_size: int = 4
    eval_batch_size: int = 0               # 0 = micro_batch_size
    gradient_accumulation_steps: int = 1
    logging_steps: int = 10
    eval_steps: int = 0                    # 0 = eval at end only
    save_steps: int = 0                    # 0 = terminal save only
    optimizer_mode: str = "auto"           # auto | flat | overlap | zero1
    comm_bucket_bytes: int = 128 << 20
    seed: int = 42
    metrics_export_address: Optional[str] = None
    uid: str = ""
    # checkpoint metadata (adapter_config.json)
    lora_r: int = 8
    lora_alpha: float = 32.0
    lora_dropout: float = 0.1
    lora_targets: tuple = ("q_proj", "v_proj")
    base_model: str = ""


class FlatAdamW:
    """Flat-buffer AdamW over the model's trainable params.

    Layout: every trainable param's storage is re-pointed into ONE flat
    bf16/fp32 buffer (`param_flat`), laid out in REVERSED registration
    order — backward produces gradients roughly front-to-back, which
    makes comm buckets contiguous. Master fp32 weights + m/v + fp32 grad
    accumulator are flat too; the fused AdamW HIP kernel updates
    master and writes the low-precision params in ONE pass, in place.

    Gradient folding happens inside `post_accumulate_grad_hook`s (fp32
    accumulation across the grad-accum window, p.grad freed
    immediately).

    Modes (cfg via ctor args):
    - "flat":   one fused all-reduce of the whole fp32 grad at step()
                (LoRA ~4.2M params: latency-bound over xGMI — optimal).
    - "overlap": buckets of `bucket_bytes` all-reduced asynchronously as
                the LAST micro-batch's backward fills them (full-param
                models: comm hides behind compute).
    - "zero1":  fp32 grad reduce-scattered across ranks; each rank runs
                AdamW on its 1/world shard of master/m/v, then the bf16
                params are all-gathered (DeepSpeed-ZeRO-1-style optimizer
                state sharding for the 13B full-param config,
                SURVEY.md §2.2).
    """

    def __init__(self, named_params, device, cfg: TrainerConfig,
                 sync: Optional[GradSynchronizer] = None,
                 mode: str = "flat", bucket_bytes: int = 128 << 20,
                 world_size: int = 1):
        import torch.distributed as dist
        self.cfg = cfg
        self.sync = sync
        self.mode = mode if (world_size > 1 or mode == "flat") else "flat"
        self.world = world_size
        self.named = list(named_params)           # registration order
        self.named.reverse()                      # ≈ backward order
        self.params = [p for _, p in self.named]
        if not self.params:
            raise ValueError("no trainable parameters")
        self.offsets = []
        off = 0
        for p in self.params:
            self.offsets.append(off)
            off += p.numel()
        # pad so the fused AdamW kernel can use 4-wide vectors and the
        # zero1 shard divides evenly
        align = 64 * max(1, world_size)
        self.numel = (off + align - 1) // align * align
        pdtype = self.params[0].dtype
        self.param_flat = torch.zeros(self.numel, dtype=pdtype,
                                      device=device)
        with torch.no_grad():
            for p, o in zip(self.params, self.offsets):
                self.param_flat[o:o + p.numel()].copy_(
                    p.detach().reshape(-1))
                p.data = self.param_flat[o:o + p.numel()].view_as(p)
        self.grad = torch.zeros(self.numel, dtype=torch.float32,
                                device=device)
        if self.mode == "zero1":
            self.shard_n = self.numel // world_size
            self.shard_off = self.shard_n * dist.get_rank() \
                if dist.is_initialized() else 0
            mslice = self.param_flat[
                self.shard_off:self.shard_off + self.shard_n]
        else:
            self.shard_n, self.shard_off = self.numel, 0
            mslice = self.param_flat
        self.master = mslice.float()
        self.m = torch.zeros_like(self.master)
        self.v = torch.zeros_like(self.master)
        self.t = 0
        # ---- grad-fold hooks + overlap buckets
        self._final_micro = False
        self._handles = []
        self._bucket_of = {}
        if self.mode == "overlap":
            elems = max(1, bucket_bytes // 4)
            buckets = []      # list of [start, end, n_params, seen]
            for p, o in zip(self.params, self.offsets):
                if not buckets or o + p.numel() - buckets[-1][0] > elems:
                    buckets.append([o, o + p.numel(), 0, 0])
                else:
                    buckets[-1][1] = o + p.numel()
                buckets[-1][2] += 1
                self._bucket_of[p] = buckets[-1]
            buckets[-1][1] = self.numel
            self.buckets = buckets
        self._install_hooks()

    def _install_hooks(self):
        for p, o in zip(self.params, self.offsets):
            n = p.numel()
            dst = self.grad[o:o + n]

            def hook(param, dst=dst, n=n):
                g = param.grad
                if g is None:
                    return
                dst.add_(g.detach().reshape(-1))
                param.grad = None
                if self._final_micro and self.mode == "overlap":
                    b = self._bucket_of[param]
                    b[3] += 1
                    if b[3] == b[2]:
                        self._launch_bucket(b)

            p.register_post_accumulate_grad_hook(hook)

    def _launch_bucket(self, b):
        import torch.distributed as dist
        b[3] = 0
        if self.world <= 1 or not dist.is_initialized():
            return
        chunk = self.grad[b[0]:b[1]]
        self._handles.append(
            dist.all_reduce(chunk, op=dist.ReduceOp.AVG, async_op=True))

    def mark_final_microbatch(self):
        """Call before the LAST micro-batch's backward of the window so
        overlap-mode buckets all-reduce as backward fills them."""
        self._final_micro = True

    def accumulate_grads_(self):
        """Kept for API compatibility — folding is done by hooks."""

    def _sync_grad(self):
        import torch.distributed as dist
        if self.world <= 1 or not dist.is_initialized():
            return
        if self.mode == "flat":
            if self.sync is not None:
                self.sync.allreduce_flat_(self.grad)
            else:
                dist.all_reduce(self.grad, op=dist.ReduceOp.AVG)
        elif self.mode == "overlap":
            # launch any bucket whose params did not all receive grads
            # this step (e.g. frozen/unused params): hooks only fire the
            # bucket when every member saw a grad, so a partial bucket
            # would otherwise never be synchronized.
            for b in self.buckets:
                if b[3] > 0:
                    self._launch_bucket(b)
            for h in self._handles:
                h.wait()
            self._handles.clear()
        elif self.mode == "zero1":
            shard = torch.empty(self.shard_n, dtype=torch.float32,
                                device=self.grad.device)
            if dist.get_backend() == "gloo":
                # gloo lacks reduce_scatter_tensor: all-reduce then slice
                dist.all_reduce(self.grad, op=dist.ReduceOp.AVG)
                shard.copy_(self.grad[self.shard_off:
                                      self.shard_off + self.shard_n])
            else:
                dist.reduce_scatter_tensor(shard, self.grad,
                                           op=dist.ReduceOp.AVG)
            self._grad_shard = shard

    def _grad_sq_sum(self) -> torch.Tensor:
        import torch.distributed as dist
        if self.mode == "zero1":
            sq = self._grad_shard.float().pow(2).sum()
            if dist.is_initialized():
                dist.all_reduce(sq, op=dist.ReduceOp.SUM)
            return sq
        return self.grad.pow(2).sum()

    def step(self, lr: float) -> float:
        import torch.distributed as dist
        self._sync_grad()
        self._final_micro = False
        gnorm = float(self._grad_sq_sum().sqrt())
        g = self._grad_shard if self.mode == "zero1" else self.grad
        max_norm = self.cfg.max_grad_norm
        if max_norm and max_norm > 0:
            scale = max_norm / (gnorm + 1e-6)
            if scale < 1.0:
                g.mul_(scale)
        self.t += 1
        pslice = self.param_flat[self.shard_off:
                                 self.shard_off + self.shard_n]
        ops.adamw_step(pslice, self.master, g, self.m, self.v,
                       lr, self.cfg.betas[0], self.cfg.betas[1],
                       self.cfg.eps, self.cfg.weight_decay, self.t)
        if self.mode == "zero1" and dist.is_initialized():
            if dist.get_backend() == "gloo":
                shards = [torch.empty_like(pslice)
                          for _ in range(self.world)]
                dist.all_gather(shards, pslice.contiguous())
                for i, s in enumerate(shards):
                    self.param_flat[i * self.shard_n:
                                    (i + 1) * self.shard_n].copy_(s)
            else:
                dist.all_gather_into_tensor(self.param_flat,
                                            pslice.contiguous())
        self.grad.zero_()
        return gnorm

    def _gather_params(self):
        """zero1: broadcast every rank's param shard into the full
        param_flat (same collective as step()'s post-update gather)."""
        import torch.distributed as dist
        if self.mode != "zero1" or self.world <= 1 \
                or not dist.is_initialized():
            return
        pslice = self.param_flat[self.shard_off:
                                 self.shard_off + self.shard_n]
        if dist.get_backend() == "gloo":
            shards = [torch.empty_like(pslice) for _ in range(self.world)]
            dist.all_gather(shards, pslice.contiguous())
            for i, sh in enumerate(shards):
                self.param_flat[i * self.shard_n:
                                (i + 1) * self.shard_n].copy_(sh)
        else:
            dist.all_gather_into_tensor(self.param_flat,
                                        pslice.contiguous())

    def state_dict(self):
        # zero1: master/m/v are the LOCAL shard — every rank must save
        # (and reload) its own state_dict.
        return {"master": self.master, "m": self.m, "v": self.v,
                "t": self.t, "mode": self.mode, "world": self.world,
                "shard_off": self.shard_off, "shard_n": self.shard_n}

    def load_state_dict(self, sd):
        if sd.get("mode", "flat") != self.mode:
            raise ValueError(
                f"optimizer mode mismatch: checkpoint {sd.get('mode')} "
                f"vs current {self.mode}")
        if self.mode == "zero1" and sd.get("world", self.world) != self.world:
            raise ValueError(
                "zero1 resume requires the same world size as the "
                f"checkpoint (ckpt {sd.get('world')} vs {self.world})")
        self.master.copy_(sd["master"])
        self.m.copy_(sd["m"])
        self.v.copy_(sd["v"])
        self.t = sd["t"]
        with torch.no_grad():
            self.param_flat[self.shard_off:
                            self.shard_off + self.shard_n].copy_(
                self.master.to(self.param_flat.dtype))
            # every rank restored only its local shard: gather so the
            # first post-resume forward sees the full restored params
            # (ADVICE r1 medium).
            self._gather_params()


def lr_at(step: int, total: int, base_lr: float, warmup_ratio: float,
          kind: str) -> float:
    warm = max(1, int(total * warmup_ratio))
    if step < warm:
        return base_lr * (step + 1) / warm
    frac = (step - warm) / max(1, total - warm)
    if kind == "cosine":
        return base_lr * 0.5 * (1.0 + math.cos(math.pi * min(1.0, frac)))
    if kind == "linear":
        return base_lr * max(0.0, 1.0 - frac)
    return base_lr


class SFTTrainer:
    def __init__(self, model, train_dataset, cfg: TrainerConfig,
                 eval_dataset=None, device=None, rank: int = 0,
                 world_size: int = 1, pad_token_id: int = 0):
        self.model = model
        self.cfg = cfg
        self.rank, self.world = rank, world_size
        self.device = device or next(model.parameters()).device
        self.train_loader = ShardedLoader(
            train_dataset, cfg.micro_batch_size, rank, world_size,
            seed=cfg.seed, pad_token_id=pad_token_id, device=self.device)
        self.eval_dataset = eval_dataset
        self.pad_token_id = pad_token_id
        sync = GradSynchronizer(world_size) if world_size > 1 else None
        named = model.trainable_parameters()
        mode = cfg.optimizer_mode
        if mode not in ("auto", "flat", "overlap", "zero1"):
            raise ValueError(f"optimizer_mode {mode!r} (want auto | "
                             f"flat | overlap | zero1)")
        if mode == "auto":
            # LoRA-size trainables (≤64M params): one fused latency-bound
            # all-reduce. Full-param models: overlap buckets with backward.
            n_train = sum(p.numel() for _, p in named)
            mode = "flat" if n_train <= (64 << 20) else "overlap"
        self.opt = FlatAdamW(named, self.device, cfg, sync, mode=mode,
                             bucket_bytes=cfg.comm_bucket_bytes,
                             world_size=world_size)
        steps_per_epoch = max(1, len(train_dataset) //
                              (cfg.micro_batch_size * world_size *
                               cfg.gradient_accumulation_steps))
        self.total_steps = (cfg.max_steps if cfg.max_steps > 0 else
                            int(steps_per_epoch * cfg.num_train_epochs))
        self.global_step = 0
        self._exporter = None
        if cfg.metrics_export_address and is_main():
            from ..metrics.remote_write import RemoteWriteExporter
            self._exporter = RemoteWriteExporter(cfg.metrics_export_address,
                                                 cfg.uid)
        self._t0 = None
        self.last_train_loss = float("nan")

    # ------------------------------------------------------------ logging
    def _log(self, record: dict, kind: str = "trainer"):
        if not is_main():
            return
        watch = os.path.join(self.cfg.output_dir, "watch")
        os.makedirs(watch, exist_ok=True)
        # open-append-close per write: a held handle leaks one fd per
        # trainer in long-lived processes (and writes are infrequent —
        # every logging_steps)
        with open(os.path.join(watch, f"{kind}_log.jsonl"), "a") as f:
            f.write(json.dumps(record) + "\n")
        if self._exporter is not None:
            if kind == "trainer":
                self._exporter.export_train_metrics(record)
            else:
                self._exporter.export_eval_metrics(record)

    # ------------------------------------------------------------- train
    def _micro_loss(self, mb):
        """One micro-batch's loss tensor (overridden by DPOTrainer)."""
        return self.model(mb["input_ids"], labels=mb["labels"])

    def _extra_log_metrics(self) -> dict:
        """Stage-specific additions to the periodic train log."""
        return {}

    def train_step(self, micro_batches) -> float:
        """One optimizer step over `gradient_accumulation_steps` micro
        batches; returns the (local) mean loss."""
        cfg = self.cfg
        acc = cfg.gradient_accumulation_steps
        total = 0.0
        for i, mb in enumerate(micro_batches):
            if i == len(micro_batches) - 1:
                self.opt.mark_final_microbatch()
            loss = self._micro_loss(mb)
            (loss / acc).backward()
            total += float(loss.detach())
        lr = lr_at(self.global_step, self.total_steps, cfg.learning_rate,
                   cfg.warmup_ratio, cfg.lr_scheduler_type)
        self.opt.step(lr)
        self.global_step += 1
        return total / acc

    def train(self):
        cfg = self.cfg
        self.model.train()
        self._t0 = time.time()
        it = iter(self.train_loader)
        # after a checkpoint resume, fast-forward the deterministic
        # loader so step N+1 sees the batches it would have seen in an
        # uninterrupted run
        for _ in range(self.global_step * cfg.gradient_accumulation_steps):
            next(it)
        while self.global_step < self.total_steps:
            mbs = [next(it) for _ in range(cfg.gradient_accumulation_steps)]
            loss = self.train_step(mbs)
            self.last_train_loss = loss
            step = self.global_step
            if cfg.logging_steps and step % cfg.logging_steps == 0:
                lr = lr_at(step - 1, self.total_steps, cfg.learning_rate,
                           cfg.warmup_ratio, cfg.lr_scheduler_type)
                loss_g = sync_scalar_mean(loss, self.device)
                elapsed = time.time() - self._t0
                self._log({
                    "current_steps": step, "total_steps": self.total_steps,
                    "loss": round(loss_g, 6), "learning_rate": lr,
                    **self._extra_log_metrics(),
                    "epoch": round(step / max(1, self.total_steps) *
                                   float(cfg.num_train_epochs), 4),
                    "percentage": round(100 * step / self.total_steps, 2),
                    "elapsed_time": round(elapsed, 2),
                    "remaining_time": round(
                        elapsed / step * (self.total_steps - step), 2),
                })
            if cfg.eval_steps and step % cfg.eval_steps == 0 and \
                    self.eval_dataset is not None:
                self.evaluate()
                self.model.train()
            if cfg.save_steps and step % cfg.save_steps == 0:
                ckpt = os.path.join(cfg.output_dir, f"checkpoint-{step}")
                if is_main():
                    self.save_checkpoint(ckpt)
                elif self.opt.mode == "zero1":
                    self.save_checkpoint_sharded(ckpt)
                if self.world > 1:
                    import torch.distributed as dist
                    if dist.is_initialized():
                        dist.barrier()
        if self.eval_dataset is not None:
            self.evaluate()
        return self.last_train_loss

    # -------------------------------------------------------------- eval
    @torch.no_grad()
    def evaluate(self) -> dict:
        self.model.eval()
        ebs = self.cfg.eval_batch_size or self.cfg.micro_batch_size
        loader = ShardedLoader(self.eval_dataset, ebs,
                               self.rank, self.world, seed=0,
                               pad_token_id=self.pad_token_id,
                               device=self.device)
        losses = []
        for mb in loader.epoch(0):
            losses.append(float(self.model(mb["input_ids"],
                                           labels=mb["labels"])))
        local = sum(losses) / max(1, len(losses))
        mean = sync_scalar_mean(local, self.device)
        metrics = {"eval_loss": round(mean, 6),
                   "eval_perplexity": round(math.exp(min(mean, 30.0)), 6),
                   "current_steps": self.global_step,
                   "total_steps": self.total_steps}
        self._log(metrics, kind="eval")
        return metrics

    # ------------------------------------------------------------ predict
    @torch.no_grad()
    def predict(self, dataset, max_new_tokens: int = 64,
                out_file: Optional[str] = None,
                eos_token_id: Optional[int] = 2):
        """Greedy generation from each example's prompt (the -100-masked
        prefix), mirroring the reference's prediction_step +
        save_predictions jsonl (cmd/tuning/trainer.py:405-489).
        KV-cache decode (prefill once, one token per forward) and stops
        at eos; eos_token_id=None disables early stop."""
        from ..data.dataset import IGNORE_INDEX
        from ..serve.engine import KVCache
        self.model.eval()
        is_llama = hasattr(self.model.cfg, "num_key_value_heads")
        results = []
        for i in range(len(dataset)):
            ex = dataset[i]
            ids, labels = list(ex["input_ids"]), list(ex["labels"])
            plen = 0
            for lb in labels:
                if lb == IGNORE_INDEX:
                    plen += 1
                else:
                    break
            prompt = ids[:plen] if plen > 0 else ids
            cur = torch.tensor([prompt], dtype=torch.long,
                               device=self.device)
            out = []
            with torch.no_grad():
                if is_llama:
                    cfg = self.model.cfg
                    max_s = min(cfg.max_position_embeddings,
                                len(prompt) + max_new_tokens + 1)
                    caches = [KVCache(
                        1, cfg.num_key_value_heads, max_s, cfg.head_dim,
                        self.device, next(self.model.parameters()).dtype)
                        for _ in range(cfg.num_hidden_layers)]
                    pos = 0
                    for _ in range(max_new_tokens):
                        if pos + cur.shape[1] > max_s:
                            break
                        logits = self.model(cur, pos0=pos,
                                            kv_caches=caches)
                        pos += cur.shape[1]
                        nxt = int(logits[0, -1].argmax())
                        if nxt == eos_token_id:
                            break
                        out.append(nxt)
                        cur = torch.tensor([[nxt]], dtype=torch.long,
                                           device=self.device)
                else:
                    for _ in range(max_new_tokens):
                        logits = self.model(cur)
                        nxt = int(logits[0, -1].argmax())
                        if nxt == eos_token_id:
                            break
                        out.append(nxt)
                        cur = torch.cat(
                            [cur, torch.tensor(
                                [[nxt]], dtype=torch.long,
                                device=self.device)], dim=1)
            results.append({
                "prompt_ids": prompt,
                "predict_ids": out,
                "label_ids": [lb for lb in labels if lb != IGNORE_INDEX],
            })
        if out_file is None:
            out_file = os.path.join(self.cfg.output_dir,
                                    "generated_predictions.jsonl")
        from .gen_metrics import generation_metrics
        metrics = generation_metrics(
            [r["predict_ids"] for r in results],
            [r["label_ids"] for r in results])
        if is_main():
            os.makedirs(os.path.dirname(out_file) or ".", exist_ok=True)
            with open(out_file, "w") as f:
                for r in results:
                    f.write(json.dumps(r) + "\n")
            self._log({**metrics, "current_steps": self.global_step,
                       "total_steps": self.total_steps}, kind="eval")
        return results

    # -------------------------------------------------------- checkpoint
    def save_checkpoint(self, out_dir: str):
        """HF-adapter layout for LoRA models; full trainable state dict
        otherwise; plus trainer resume state."""
        os.makedirs(out_dir, exist_ok=True)
        has_lora = any(isinstance(m, LoRALinearModule)
                       for m in self.model.modules())
        if has_lora:
            save_adapter(self.model, out_dir, r=self.cfg.lora_r,
                         alpha=self.cfg.lora_alpha,
                         dropout=self.cfg.lora_dropout,
                         target_modules=list(self.cfg.lora_targets),
                         base_model_name_or_path=self.cfg.base_model)
        else:
            from ..models import LlamaForCausalLM
            if isinstance(self.model, LlamaForCausalLM):
                # full-param Llama: HF-format export (config.json +
                # HF-named safetensors) so transformers / the hf_io
                # loader can consume the checkpoint directly
                from ..models.hf_io import save_hf_model
                save_hf_model(self.model, out_dir)
            else:
                from safetensors.torch import save_file
                sd = {n: p.detach().cpu().contiguous()
                      for n, p in self.model.trainable_parameters()}
                save_file(sd, os.path.join(out_dir, "model.safetensors"))
        torch.save({"optimizer": self.opt.state_dict(),
                    "global_step": self.global_step},
                   os.path.join(out_dir, self._opt_state_name(self.rank)))
        return out_dir

    @staticmethod
    def _opt_state_name(rank: int) -> str:
        return ("trainer_state.pt" if rank == 0
                else f"trainer_state_rank{rank}.pt")

    def save_checkpoint_sharded(self, out_dir: str):
        """zero1 companion to save_checkpoint for ranks != 0: write this
        rank's optimizer shard next to rank 0's checkpoint."""
        os.makedirs(out_dir, exist_ok=True)
        torch.save({"optimizer": self.opt.state_dict(),
                    "global_step": self.global_step},
                   os.path.join(out_dir, self._opt_state_name(self.rank)))

    def load_checkpoint(self, ckpt_dir: str):
        path = os.path.join(ckpt_dir, self._opt_state_name(self.rank))
        if not os.path.exists(path) and self.rank != 0:
            # pre-sharding checkpoint (or flat mode saved by rank 0 only)
            path = os.path.join(ckpt_dir, "trainer_state.pt")
        state = torch.load(path, map_location=self.device,
                           weights_only=False)
        self.opt.load_state_dict(state["optimizer"])
        self.global_step = state["global_step"]


# ------------------------------------------------------------------ DPO
class _AdaptersDisabled:
    """Zero every LoRA module's scale so the model computes EXACTLY the
    frozen base — the DPO reference policy without a second weight copy
    (the adapters start at B=0, so base == pre-DPO policy). The fused
    QKVProj path reads the same module `scale` attribute, so both
    dispatch paths are covered. Not thread-safe across concurrent
    trainers on one model (the trainer owns its model)."""

    def __init__(self, model):
        self.mods = [m for m in model.modules()
                     if isinstance(m, LoRALinearModule)]

    def __enter__(self):
        self.saved = [m.scale for m in self.mods]
        for m in self.mods:
            m.scale = 0.0
        return self

    def __exit__(self, *exc):
        for m, s in zip(self.mods, self.saved):
            m.scale = s
        return False


class DPOTrainer(SFTTrainer):
    """Direct Preference Optimization over (chosen, rejected) pairs —
    the reference's declared-but-unimplemented stage=dpo
    (cmd/tuning/parser.py:112-221) made real. LoRA-only: the reference
    policy is the model with adapters disabled (exact, free — no second
    model in memory). Loss per pair:
        -logsigmoid(beta * ((pol_c - ref_c) - (pol_r - ref_r)))
    where each term is the sum log-probability of that completion's
    non-masked tokens (model.sequence_logprobs -> the fused masked
    xent kernels, unreduced)."""

    def __init__(self, model, train_dataset, cfg: TrainerConfig,
                 device=None, rank: int = 0, world_size: int = 1,
                 eval_dataset=None, pad_token_id: int = 0,
                 beta: float = 0.1):
        from ..data.preference import collate_preference
        if not any(isinstance(m, LoRALinearModule)
                   for m in model.modules()):
            raise ValueError("DPO requires LoRA adapters (the frozen "
                             "base is the reference policy)")
        super().__init__(model, train_dataset, cfg, device=device,
                         rank=rank, world_size=world_size,
                         eval_dataset=eval_dataset,
                         pad_token_id=pad_token_id)
        self.beta = beta
        self.train_loader.collate_fn = collate_preference
        # standard DPO practice: dropout off — the loss compares policy
        # and reference LOG-PROBABILITIES of fixed sequences, and
        # stochastic masks turn that into a noisy estimate
        for m in model.modules():
            if isinstance(m, LoRALinearModule):
                m.dropout = 0.0
        self.last_margin = 0.0
        self.last_acc = 0.0

    def _extra_log_metrics(self) -> dict:
        return {"reward_margin": round(self.last_margin, 6),
                "pref_accuracy": round(self.last_acc, 4)}

    def _micro_loss(self, mb):
        ids = torch.cat([mb["chosen_ids"], mb["rejected_ids"]], dim=0) \
            if mb["chosen_ids"].shape[1] == mb["rejected_ids"].shape[1] \
            else None
        with torch.no_grad(), _AdaptersDisabled(self.model):
            if ids is not None:
                labels = torch.cat([mb["chosen_labels"],
                                    mb["rejected_labels"]], dim=0)
                ref = self.model.sequence_logprobs(ids, labels)
                B = mb["chosen_ids"].shape[0]
                ref_c, ref_r = ref[:B], ref[B:]
            else:              # chosen/rejected padded to different S
                ref_c = self.model.sequence_logprobs(
                    mb["chosen_ids"], mb["chosen_labels"])
                ref_r = self.model.sequence_logprobs(
                    mb["rejected_ids"], mb["rejected_labels"])
        if ids is not None:
            labels = torch.cat([mb["chosen_labels"],
                                mb["rejected_labels"]], dim=0)
            pol = self.model.sequence_logprobs(ids, labels)
            B = mb["chosen_ids"].shape[0]
            pol_c, pol_r = pol[:B], pol[B:]
        else:
            pol_c = self.model.sequence_logprobs(mb["chosen_ids"],
                                                 mb["chosen_labels"])
            pol_r = self.model.sequence_logprobs(mb["rejected_ids"],
                                                 mb["rejected_labels"])
        margin = (pol_c - ref_c) - (pol_r - ref_r)
        self.last_margin = float(margin.detach().mean())
        self.last_acc = float((margin.detach() > 0).float().mean())
        return -torch.nn.functional.logsigmoid(
            self.beta * margin.float()).mean()

    def evaluate(self) -> dict:
        """Preference accuracy + margin on the eval set (the SFT
        eval_loss contract does not apply to pairs)."""
        from ..data.preference import collate_preference
        self.model.eval()
        ebs = self.cfg.eval_batch_size or self.cfg.micro_batch_size
        loader = ShardedLoader(self.eval_dataset, ebs,
                               self.rank, self.world, seed=0,
                               pad_token_id=self.pad_token_id,
                               device=self.device,
                               collate_fn=collate_preference)
        margins = []
        with torch.no_grad():
            for mb in loader.epoch(0):
                with _AdaptersDisabled(self.model):
                    rc = self.model.sequence_logprobs(
                        mb["chosen_ids"], mb["chosen_labels"])
                    rr = self.model.sequence_logprobs(
                        mb["rejected_ids"], mb["rejected_labels"])
                pc = self.model.sequence_logprobs(
                    mb["chosen_ids"], mb["chosen_labels"])
                pr = self.model.sequence_logprobs(
                    mb["rejected_ids"], mb["rejected_labels"])
                margins.append(((pc - rc) - (pr - rr)))
        m = torch.cat(margins) if margins else torch.zeros(1)
        metrics = {"eval_margin": round(float(m.mean()), 6),
                   "eval_pref_accuracy": round(
                       float((m > 0).float().mean()), 6),
                   "current_steps": self.global_step,
                   "total_steps": self.total_steps}
        self._log(metrics, kind="eval")
        self.model.train()
        return metrics
