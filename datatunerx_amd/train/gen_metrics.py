"""Generation-eval metrics (the reference's GenEvalSeq2SeqTrainer path:
cmd/tuning/trainer.py:29-172 computes rouge/bleu over generated
predictions; callback.py logs the keys). Pure-python token-level
implementations (no network for nltk/rouge packages)."""

from __future__ import annotations

from collections import Counter
from typing import Dict, List, Sequence


def _lcs_len(a: Sequence, b: Sequence) -> int:
    if not a or not b:
        return 0
    prev = [0] * (len(b) + 1)
    for x in a:
        cur = [0]
        for j, y in enumerate(b, 1):
            cur.append(prev[j - 1] + 1 if x == y
                       else max(prev[j], cur[-1]))
        prev = cur
    return prev[-1]


def rouge_l(pred: Sequence, ref: Sequence) -> float:
    """token-level ROUGE-L F1."""
    lcs = _lcs_len(pred, ref)
    if lcs == 0:
        return 0.0
    p = lcs / len(pred)
    r = lcs / len(ref)
    return 2 * p * r / (p + r)


def bleu(pred: Sequence, ref: Sequence, max_n: int = 4) -> float:
    """token-level BLEU with uniform n-gram weights + brevity penalty."""
    import math
    if not pred or not ref:
        return 0.0
    logs = []
    for n in range(1, max_n + 1):
        pn = [tuple(pred[i:i + n]) for i in range(len(pred) - n + 1)]
        rn = Counter(tuple(ref[i:i + n]) for i in range(len(ref) - n + 1))
        if not pn:
            logs.append(math.log(1e-9))
            continue
        hit = 0
        used = Counter()
        for g in pn:
            if used[g] < rn.get(g, 0):
                hit += 1
                used[g] += 1
        logs.append(math.log(max(hit / len(pn), 1e-9)))
    bp = 1.0 if len(pred) >= len(ref) else \
        math.exp(1.0 - len(ref) / max(1, len(pred)))
    import math as _m
    return bp * _m.exp(sum(logs) / max_n)


def generation_metrics(preds: List[Sequence],
                       refs: List[Sequence]) -> Dict[str, float]:
    """Mean rouge-l / bleu over a prediction set (keys match the
    reference callback's eval metric names)."""
    if not preds:
        return {"predict_rouge-l": 0.0, "predict_bleu-4": 0.0}
    rl = sum(rouge_l(p, r) for p, r in zip(preds, refs)) / len(preds)
    bl = sum(bleu(p, r) for p, r in zip(preds, refs)) / len(preds)
    return {"predict_rouge-l": round(rl, 6), "predict_bleu-4": round(bl, 6)}
