"""Dataset ingest + supervised preprocessing.

Re-implements the reference's Ray-Data pipeline (cmd/tuning/train.py:
58-135, 339-351) natively: CSV ingest with feature→column mapping
(Dataset CR `features[].mapTo` — finetune_controller.go:655-680),
template encode, source masking with IGNORE_INDEX=-100, proportional
truncation to cutoff_len (train.py:88-111), per-rank strided sharding
(replaces `ray.train.get_dataset_shard`).
"""

from __future__ import annotations

import csv
from dataclasses import dataclass
from typing import Dict, Iterator, List, Optional

import torch

from .templates import get_template

IGNORE_INDEX = -100
DEFAULT_CUTOFF_LEN = 1024   # train.py:49-51


class ByteTokenizer:
    """Self-contained byte-level tokenizer for tests/synthetic runs (no
    network for real tokenizer files). ids: 0=pad, 1=bos, 2=eos,
    3..258=bytes."""

    def __init__(self):
        self.pad_token_id = 0
        self.bos_token_id = 1
        self.eos_token_id = 2
        self.vocab_size = 259

    def encode(self, text: str, add_special_tokens: bool = False):
        ids = [3 + b for b in text.encode("utf-8")]
        if add_special_tokens:
            ids = [self.bos_token_id] + ids
        return ids

    def decode(self, ids):
        # ids past the byte range (models with a larger vocab emit
        # them on random weights) are dropped, not an error
        return bytes(i - 3 for i in ids
                     if 3 <= i < 259).decode("utf-8", errors="replace")


def _map_row(raw: dict, column_map: Optional[Dict[str, str]]):
    if column_map:
        return {feat: str(raw.get(col, "") or "")
                for feat, col in column_map.items()}
    return {k: str(v) if v is not None else "" for k, v in raw.items()}


def read_rows(path: str, column_map: Optional[Dict[str, str]] = None
              ) -> List[Dict[str, str]]:
    """Rows from a .csv, .json (list of objects) or .jsonl file.
    column_map: {"instruction": <source col>, ...} (the Dataset CR
    feature mapping). The reference ingests CSV only (Ray Data,
    train.py:339-351); json/jsonl cover the common alpaca-style
    instruction files."""
    import json as _json
    rows = []
    if path.endswith(".jsonl"):
        with open(path, encoding="utf-8-sig") as f:
            for line in f:
                line = line.strip()
                if line:
                    rows.append(_map_row(_json.loads(line), column_map))
        return rows
    if path.endswith(".json"):
        with open(path, encoding="utf-8-sig") as f:
            data = _json.load(f)
        if not isinstance(data, list):
            raise ValueError(f"{path}: expected a JSON list of objects")
        return [_map_row(r, column_map) for r in data]
    # default: CSV. utf-8-sig strips the BOM Excel prepends (a plain
    # utf-8 read leaks \ufeff into the first header and breaks mapping)
    with open(path, newline="", encoding="utf-8-sig") as f:
        for raw in csv.DictReader(f):
            rows.append(_map_row(raw, column_map))
    return rows


# back-compat name (CSV was the only format through round 2)
read_csv_rows = read_rows


def preprocess_supervised_example(tokenizer, template_name: str,
                                  instruction: str, response: str,
                                  history=None, system: str = "",
                                  cutoff_len: int = DEFAULT_CUTOFF_LEN):
    """-> (input_ids, labels) with source positions = IGNORE_INDEX and
    proportional truncation (train.py:73-117 semantics): per turn,
    max_source_len = cutoff * len(src)/(len(src)+len(tgt))."""
    template = get_template(template_name)
    pairs = template.encode_multiturn(tokenizer, instruction, response,
                                      history, system)
    input_ids: List[int] = []
    labels: List[int] = []
    for src, tgt in pairs:
        total = len(src) + len(tgt)
        if total > cutoff_len:
            max_src = int(cutoff_len * (len(src) / total))
            max_tgt = cutoff_len - max_src
            src, tgt = src[:max_src], tgt[:max_tgt]
        if len(input_ids) + len(src) + len(tgt) > cutoff_len:
            break
        input_ids.extend(src + tgt)
        labels.extend([IGNORE_INDEX] * len(src) + tgt)
    input_ids = input_ids[:cutoff_len]
    labels = labels[:cutoff_len]
    return input_ids, labels


@dataclass
class SFTDataset:
    """Tokenized supervised dataset held in memory (tensorized lazily)."""
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
            ids, labels = preprocess_supervised_example(
                tokenizer, template_name, row.get("instruction", ""),
                row.get("response", ""), system=system, cutoff_len=cutoff_len)
            if any(l != IGNORE_INDEX for l in labels):
                ex.append({"input_ids": ids, "labels": labels})
        return cls(ex)

    @classmethod
    def from_csv(cls, path: str, tokenizer, column_map=None,
                 template_name: str = "llama2",
                 cutoff_len: int = DEFAULT_CUTOFF_LEN):
        """Accepts .csv, .json (list) or .jsonl (see read_rows)."""
        return cls.from_rows(read_rows(path, column_map), tokenizer,
                             template_name, cutoff_len)

    from_file = from_csv

    @classmethod
    def from_rows_pt(cls, rows, tokenizer,
                     cutoff_len: int = DEFAULT_CUTOFF_LEN):
        """Pretraining (stage=pt, parser.py:112-221 stage field): pack
        raw text into fixed cutoff_len blocks with eos separators and
        train on EVERY position (labels = input_ids, no -100 prompt
        masking). Rows use the same columns as sft; instruction and
        response are concatenated as plain text."""
        eos = ([tokenizer.eos_token_id]
               if getattr(tokenizer, "eos_token_id", None) is not None
               else [])
        stream: List[int] = []
        for row in rows:
            text = " ".join(s for s in (row.get("instruction", ""),
                                        row.get("response", "")) if s)
            if not text:
                continue
            try:
                ids = tokenizer.encode(text, add_special_tokens=False)
            except TypeError:
                ids = tokenizer.encode(text)
            stream.extend(ids + eos)
        ex = []
        for i in range(0, len(stream) - cutoff_len + 1, cutoff_len):
            blk = stream[i:i + cutoff_len]
            ex.append({"input_ids": blk, "labels": list(blk)})
        if not ex and stream:            # corpus shorter than one block
            ex.append({"input_ids": stream, "labels": list(stream)})
        return cls(ex)

    @classmethod
    def synthetic(cls, n_examples: int, seq_len: int, vocab_size: int,
                  seed: int = 0, mask_frac: float = 0.25):
        """Fixed-shape synthetic instruction data for benches (BASELINE
        contract: synthetic data, stated in bench output)."""
        g = torch.Generator().manual_seed(seed)
        ex = []
        n_src = max(1, int(seq_len * mask_frac)) if mask_frac > 0 else 0
        for _ in range(n_examples):
            ids = torch.randint(3, vocab_size, (seq_len,), generator=g)
            labels = ids.clone()
            labels[:n_src] = IGNORE_INDEX
            ex.append({"input_ids": ids.tolist(), "labels": labels.tolist()})
        return cls(ex)


def collate(batch, pad_token_id: int = 0, pad_to_multiple_of: int = 4,
            device=None):
    """Right-pad to the batch max (labels padded with -100); pad length
    rounded to a multiple (train.py:282-286 collator hint)."""
    maxlen = max(len(b["input_ids"]) for b in batch)
    m = pad_to_multiple_of
    maxlen = ((maxlen + m - 1) // m) * m
    ids = torch.full((len(batch), maxlen), pad_token_id, dtype=torch.long)
    labels = torch.full((len(batch), maxlen), IGNORE_INDEX, dtype=torch.long)
    for i, b in enumerate(batch):
        n = len(b["input_ids"])
        ids[i, :n] = torch.as_tensor(b["input_ids"], dtype=torch.long)
        labels[i, :n] = torch.as_tensor(b["labels"], dtype=torch.long)
    if device is not None:
        ids, labels = ids.to(device), labels.to(device)
    return {"input_ids": ids, "labels": labels}


class ShardedLoader:
    """Per-rank strided sampler over a dataset, infinite epochs with
    deterministic shuffling (replaces Ray object-store sharding —
    SURVEY.md §2.3)."""

    def __init__(self, dataset, batch_size: int, rank: int = 0,
                 world_size: int = 1, seed: int = 0, pad_token_id: int = 0,
                 device=None, drop_last: bool = True, collate_fn=None):
        self.ds, self.bs = dataset, batch_size
        self.rank, self.world = rank, world_size
        self.seed, self.pad, self.device = seed, pad_token_id, device
        self.drop_last = drop_last
        self.collate_fn = collate_fn or collate

    def epoch(self, epoch_idx: int) -> Iterator[dict]:
        g = torch.Generator().manual_seed(self.seed + epoch_idx)
        perm = torch.randperm(len(self.ds), generator=g).tolist()
        shard = perm[self.rank::self.world]
        n_full = len(shard) // self.bs
        for i in range(n_full):
            idx = shard[i * self.bs:(i + 1) * self.bs]
            yield self.collate_fn([self.ds[j] for j in idx], self.pad,
                                  device=self.device)

    def __iter__(self):
        e = 0
        while True:
            yield from self.epoch(e)
            e += 1
