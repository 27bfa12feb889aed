"""Admission validation + defaulting — the native replacement for the
reference's validating/mutating webhooks (registered at
controller_manager.go:112-135; bodies in the external meta-server
module, so the rules here are reconstructed from the fields the
controllers consume).

Every malformed shape must surface as ValidationError, never as an
uncontrolled AttributeError/TypeError — clients send arbitrary JSON
(property-fuzzed in tests/test_validation_fuzz.py)."""

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


def _dict(x, path: str, errs: List[str]) -> dict:
    """x as a dict; None/missing -> {}; anything else is an error."""
    if x is None:
        return {}
    if isinstance(x, dict):
        return x
    errs.append(f"{path} must be a map")
    return {}


def _validate_finetune_spec(spec, path: str, errs: List[str]):
    spec = _dict(spec, path, errs)
    _require(bool(spec.get("llm")), f"{path}.llm is required", errs)
    _require(bool(spec.get("dataset")), f"{path}.dataset is required", errs)
    hp = _dict(spec.get("hyperparameter"), f"{path}.hyperparameter", errs)
    _require(bool(hp.get("hyperparameterRef")),
             f"{path}.hyperparameter.hyperparameterRef is required", errs)
    node = spec.get("node", 1)
    _require(isinstance(node, int) and not isinstance(node, bool) and
             1 <= node <= 8,
             f"{path}.node must be an int in [1, 8]", errs)


def default_(obj: ApiObject) -> None:
    """Mutating-webhook parity: fill defaults in place. Non-map fields
    are left untouched for validate_ to reject."""
    if isinstance(obj, FinetuneJob):
        ft = obj.spec.get("fineTune")
        if ft is None:
            ft = obj.spec["fineTune"] = {}
        if isinstance(ft, dict):
            fs = ft.get("finetuneSpec")
            if fs is None:
                fs = ft["finetuneSpec"] = {}
            if isinstance(fs, dict):
                fs.setdefault("node", 1)
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
    name_ok = isinstance(obj.metadata.name, str) and obj.metadata.name
    _require(bool(name_ok), "metadata.name is required", errs)
    if name_ok:
        _require(_NAME_RE.match(obj.metadata.name) is not None and
                 len(obj.metadata.name) <= 253,
                 "metadata.name must be DNS-1123", errs)
    if isinstance(obj, FinetuneJob):
        ft = _dict(obj.spec.get("fineTune"), "spec.fineTune", errs)
        _validate_finetune_spec(ft.get("finetuneSpec"),
                                "spec.fineTune.finetuneSpec", errs)
        sc = _dict(obj.spec.get("serveConfig"), "spec.serveConfig", errs)
        tp = sc.get("tensorParallel", sc.get("gpus", 1))
        _require(isinstance(tp, int) and not isinstance(tp, bool) and
                 1 <= tp <= 8,
                 "spec.serveConfig.tensorParallel must be an int in "
                 "[1, 8] (one 8-GPU MI355X node)", errs)
        tmpl = sc.get("template")
        if tmpl is not None:
            from ..data.templates import TEMPLATES
            _require(tmpl in TEMPLATES,
                     f"spec.serveConfig.template must be one of "
                     f"{sorted(TEMPLATES)}", errs)
    elif isinstance(obj, Finetune):
        _validate_finetune_spec(obj.spec, "spec", errs)
    elif isinstance(obj, FinetuneExperiment):
        jobs = obj.spec.get("finetuneJobs")
        if jobs is None:
            jobs = []
        if not isinstance(jobs, list):
            errs.append("spec.finetuneJobs must be a list")
            jobs = []
        _require(len(jobs) > 0, "spec.finetuneJobs must be non-empty",
                 errs)
        seen = set()
        for i, js in enumerate(jobs):
            js = _dict(js, f"spec.finetuneJobs[{i}]", errs)
            name = js.get("name", "")
            _require(isinstance(name, str) and bool(name),
                     f"spec.finetuneJobs[{i}].name required", errs)
            _require(name not in seen,
                     f"spec.finetuneJobs[{i}].name duplicated", errs)
            seen.add(name)
            spec = _dict(js.get("spec"), f"spec.finetuneJobs[{i}].spec",
                         errs)
            ft = _dict(spec.get("fineTune"),
                       f"spec.finetuneJobs[{i}].spec.fineTune", errs)
            _validate_finetune_spec(
                ft.get("finetuneSpec"),
                f"spec.finetuneJobs[{i}].spec.fineTune.finetuneSpec",
                errs)
    elif isinstance(obj, Hyperparameter):
        params = obj.spec.get("parameters")
        _require(isinstance(params, dict),
                 "spec.parameters must be a map", errs)
        if isinstance(params, dict):
            ep = params.get("epochs")
            _require(ep is None or (isinstance(ep, int) and
                                    not isinstance(ep, bool) and ep >= 1),
                     "spec.parameters.epochs must be >= 1", errs)
            bs = params.get("batchSize")
            _require(bs is None or (isinstance(bs, int) and
                                    not isinstance(bs, bool) and bs >= 1),
                     "spec.parameters.batchSize must be >= 1", errs)
            _require(not (params.get("int4") and params.get("int8")),
                     "int4 and int8 are mutually exclusive", errs)
            st = params.get("stage")
            _require(st is None or st in ("sft", "pt", "dpo"),
                     "spec.parameters.stage must be sft, pt or dpo", errs)
    elif isinstance(obj, Dataset):
        md = _dict(obj.spec.get("datasetMetadata"),
                   "spec.datasetMetadata", errs)
        info = _dict(md.get("datasetInfo"),
                     "spec.datasetMetadata.datasetInfo", errs)
        _require(bool(info.get("subsets")),
                 "spec.datasetMetadata.datasetInfo.subsets required", errs)
        feats = info.get("features")
        if feats is None:
            feats = []
        if not isinstance(feats, list):
            errs.append("spec.datasetMetadata.datasetInfo.features "
                        "must be a list")
            feats = []
        for i, f in enumerate(feats):
            f = _dict(f, f"features[{i}]", errs)
            # instruction/response for sft+pt; chosen/rejected for dpo
            _require(f.get("name") in ("instruction", "response",
                                       "chosen", "rejected"),
                     f"features[{i}].name must be instruction|response|"
                     f"chosen|rejected", errs)
    elif isinstance(obj, LLM):
        pass                                     # spec opaque
    if errs:
        raise ValidationError(
            f"{obj.kind}/{obj.metadata.name}: " + "; ".join(errs))
