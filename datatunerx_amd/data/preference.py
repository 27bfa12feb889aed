"""Preference datasets for DPO (stage=dpo).

The reference declares stage rm/ppo/dpo in its parser
(cmd/tuning/parser.py:112-221) but only ever runs sft; this module plus
train/trainer.py's DPOTrainer make dpo real. Rows carry
(instruction, chosen, rejected); both completions encode through the
same chat template + -100 prompt masking as SFT, so the per-sequence
log-probability of exactly the completion tokens falls out of the
existing masked cross-entropy kernels.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional

import torch

from .dataset import (DEFAULT_CUTOFF_LEN, IGNORE_INDEX, collate,
                      preprocess_supervised_example, read_csv_rows)


@dataclass
class PreferenceDataset:
    """Each example: chosen_ids/chosen_labels/rejected_ids/
    rejected_labels (same masking semantics as SFTDataset)."""
    examples: List[Dict[str, List[int]]]

    def __len__(self):
        return len(self.examples)

    def __getitem__(self, i):
        return self.examples[i]

    @classmethod
    def from_rows(cls, rows, tokenizer, template_name: str = "llama2",
                  cutoff_len: int = DEFAULT_CUTOFF_LEN, system: str = ""):
        ex = []
        for row in rows:
            instr = row.get("instruction", "")
            ch, rj = row.get("chosen", ""), row.get("rejected", "")
            if not ch or not rj:
                continue
            cids, clab = preprocess_supervised_example(
                tokenizer, template_name, instr, ch, system=system,
                cutoff_len=cutoff_len)
            rids, rlab = preprocess_supervised_example(
                tokenizer, template_name, instr, rj, system=system,
                cutoff_len=cutoff_len)
            if any(l != IGNORE_INDEX for l in clab) and \
                    any(l != IGNORE_INDEX for l in rlab):
                ex.append({"chosen_ids": cids, "chosen_labels": clab,
                           "rejected_ids": rids, "rejected_labels": rlab})
        return cls(ex)

    @classmethod
    def from_csv(cls, path: str, tokenizer,
                 column_map: Optional[Dict[str, str]] = None,
                 template_name: str = "llama2",
                 cutoff_len: int = DEFAULT_CUTOFF_LEN):
        cmap = dict(column_map or {})
        cmap.setdefault("instruction", "instruction")
        cmap.setdefault("chosen", "chosen")
        cmap.setdefault("rejected", "rejected")
        return cls.from_rows(read_csv_rows(path, cmap), tokenizer,
                             template_name, cutoff_len)

    @classmethod
    def synthetic(cls, n_examples: int, seq_len: int, vocab_size: int,
                  seed: int = 0, mask_frac: float = 0.25):
        """Shared masked prompt, two different completions per example."""
        g = torch.Generator().manual_seed(seed)
        n_src = max(1, int(seq_len * mask_frac))
        ex = []
        for _ in range(n_examples):
            prompt = torch.randint(3, vocab_size, (n_src,), generator=g)
            ch = torch.randint(3, vocab_size, (seq_len - n_src,),
                               generator=g)
            rj = torch.randint(3, vocab_size, (seq_len - n_src,),
                               generator=g)
            def pack(comp):
                ids = torch.cat([prompt, comp]).tolist()
                labels = [IGNORE_INDEX] * n_src + comp.tolist()
                return ids, labels
            cids, clab = pack(ch)
            rids, rlab = pack(rj)
            ex.append({"chosen_ids": cids, "chosen_labels": clab,
                       "rejected_ids": rids, "rejected_labels": rlab})
        return cls(ex)


def collate_preference(batch, pad_token_id: int = 0,
                       pad_to_multiple_of: int = 4, device=None):
    """Pads chosen and rejected INDEPENDENTLY (they may differ in
    length) through the standard collate; returns the four tensors."""
    ch = collate([{"input_ids": b["chosen_ids"],
                   "labels": b["chosen_labels"]} for b in batch],
                 pad_token_id, pad_to_multiple_of, device)
    rj = collate([{"input_ids": b["rejected_ids"],
                   "labels": b["rejected_labels"]} for b in batch],
                 pad_token_id, pad_to_multiple_of, device)
    return {"chosen_ids": ch["input_ids"], "chosen_labels": ch["labels"],
            "rejected_ids": rj["input_ids"],
            "rejected_labels": rj["labels"]}
