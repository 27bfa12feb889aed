"""Admission validation + defaulting — the native replacement for the
reference's validating/mutating webhooks (registered at
controller_manager.go:112-135; bodies in the external meta-server
module, so the rules here are reconstructed from the fields the
controllers consume)."""

from __future__ import annotations

import re
from typing import List

from .types import (ApiObject, Dataset, Finetune, FinetuneExperiment,
                    FinetuneJob, Hyperparameter, LLM)

_NAME_RE = re.compile(r"^[a-z0-9]([a-z0-9.-]*[a-z0-9])?$")


class ValidationError(ValueError):
    pass


def _require(cond: bool, msg: str, errs: List[str]):
    if not cond:
        errs.append(msg)


def _validate_finetune_spec(spec: dict, path: str, errs: List[str]):
    _require(bool(spec.get("llm")), f"{path}.llm is required", errs)
    _require(bool(spec.get("dataset")), f"{path}.dataset is required", errs)
    hp = spec.get("hyperparameter") or {}
    _require(bool(hp.get("hyperparameterRef")),
             f"{path}.hyperparameter.hyperparameterRef is required", errs)
    node = spec.get("node", 1)
    _require(isinstance(node, int) and 1 <= node <= 8,
             f"{path}.node must be an int in [1, 8]", errs)


def default_(obj: ApiObject) -> None:
    """Mutating-webhook parity: fill defaults in place."""
    if isinstance(obj, FinetuneJob):
        ft = obj.spec.setdefault("fineTune", {})
        ft.setdefault("finetuneSpec", {}).setdefault("node", 1)
        obj.spec.setdefault("scoringPluginConfig", {"name": "builtin"})
    elif isinstance(obj, Finetune):
        obj.spec.setdefault("node", 1)
    elif isinstance(obj, Hyperparameter):
        obj.spec.setdefault("parameters", {})


def validate_(obj: ApiObject) -> None:
    """Validating-webhook parity; raises ValidationError with every
    problem listed (webhooks for FinetuneJob, FinetuneExperiment, LLM,
    Hyperparameter, Dataset — controller_manager.go:112-135)."""
    errs: List[str] = []
    _require(bool(obj.metadata.name), "metadata.name is required", errs)
    if obj.metadata.name:
        _require(_NAME_RE.match(obj.metadata.name) is not None and
                 len(obj.metadata.name) <= 253,
                 "metadata.name must be DNS-1123", errs)
    if isinstance(obj, FinetuneJob):
        ft = (obj.spec.get("fineTune") or {}).get("finetuneSpec") or {}
        _validate_finetune_spec(ft, "spec.fineTune.finetuneSpec", errs)
        sc = obj.spec.get("serveConfig") or {}
        tp = sc.get("tensorParallel", sc.get("gpus", 1))
        _require(isinstance(tp, int) and 1 <= tp <= 8,
                 "spec.serveConfig.tensorParallel must be an int in "
                 "[1, 8] (one 8-GPU MI355X node)", errs)
    elif isinstance(obj, Finetune):
        _validate_finetune_spec(obj.spec, "spec", errs)
    elif isinstance(obj, FinetuneExperiment):
        jobs = obj.spec.get("finetuneJobs") or []
        _require(len(jobs) > 0, "spec.finetuneJobs must be non-empty",
                 errs)
        seen = set()
        for i, js in enumerate(jobs):
            name = js.get("name", "")
            _require(bool(name), f"spec.finetuneJobs[{i}].name required",
                     errs)
            _require(name not in seen,
                     f"spec.finetuneJobs[{i}].name duplicated", errs)
            seen.add(name)
            ft = ((js.get("spec") or {}).get("fineTune") or {}) \
                .get("finetuneSpec") or {}
            _validate_finetune_spec(
                ft, f"spec.finetuneJobs[{i}].spec.fineTune.finetuneSpec",
                errs)
    elif isinstance(obj, Hyperparameter):
        params = obj.spec.get("parameters")
        _require(isinstance(params, dict),
                 "spec.parameters must be a map", errs)
        if isinstance(params, dict):
            ep = params.get("epochs")
            _require(ep is None or (isinstance(ep, int) and ep >= 1),
                     "spec.parameters.epochs must be >= 1", errs)
            bs = params.get("batchSize")
            _require(bs is None or (isinstance(bs, int) and bs >= 1),
                     "spec.parameters.batchSize must be >= 1", errs)
            _require(not (params.get("int4") and params.get("int8")),
                     "int4 and int8 are mutually exclusive", errs)
            st = params.get("stage")
            _require(st is None or st in ("sft", "pt", "dpo"),
                     "spec.parameters.stage must be sft, pt or dpo", errs)
    elif isinstance(obj, Dataset):
        info = ((obj.spec.get("datasetMetadata") or {})
                .get("datasetInfo") or {})
        _require(bool(info.get("subsets")),
                 "spec.datasetMetadata.datasetInfo.subsets required", errs)
        feats = info.get("features") or []
        for i, f in enumerate(feats):
            _require(f.get("name") in ("instruction", "response"),
                     f"features[{i}].name must be instruction|response",
                     errs)
    elif isinstance(obj, LLM):
        pass                                     # spec opaque
    if errs:
        raise ValidationError(
            f"{obj.kind}/{obj.metadata.name}: " + "; ".join(errs))
