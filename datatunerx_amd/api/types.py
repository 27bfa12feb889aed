"""Declarative API objects — the same schema surface as the reference's
CRDs (groups finetune./core./extension.datatunerx.io
v1beta1), re-declared natively. Objects serialize to the standard
{apiVersion, kind, metadata, spec, status} YAML shape so manifests are
drop-in familiar; the file-backed store (store.py) plays the apiserver.

State machines (values match the reference exactly):
  Finetune.status.state: Init | Pending | Running | Successful | Failed
  FinetuneJob.status.state: Init | Finetune | BuildImage | Serve |
                            Successful | Failed
  FinetuneExperiment.status.state: Pending | Processing | Success | Failed
"""

from __future__ import annotations

import copy
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

GROUP_FINETUNE = "finetune.datatunerx.io/v1beta1"
GROUP_CORE = "core.datatunerx.io/v1beta1"
GROUP_EXTENSION = "extension.datatunerx.io/v1beta1"

FINALIZER = "finetune.datatunerx.io/finalizer"


def _now() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())


@dataclass
class ObjectMeta:
    name: str = ""
    namespace: str = "default"
    uid: str = ""
    labels: Dict[str, str] = field(default_factory=dict)
    owner_references: List[Dict[str, str]] = field(default_factory=list)
    finalizers: List[str] = field(default_factory=list)
    creation_timestamp: str = field(default_factory=_now)
    deletion_timestamp: Optional[str] = None


class ApiObject:
    """Base: kind/apiVersion + metadata + dict-backed spec/status."""

    kind = "ApiObject"
    api_version = GROUP_FINETUNE

    def __init__(self, name: str = "", namespace: str = "default",
                 spec: Optional[dict] = None, status: Optional[dict] = None,
                 metadata: Optional[ObjectMeta] = None):
        self.metadata = metadata or ObjectMeta(name=name, namespace=namespace)
        if metadata is None:
            self.metadata.uid = f"{self.kind.lower()}-{name}-{int(time.time()*1000)%10**10}"
        self.spec: Dict[str, Any] = spec or {}
        self.status: Dict[str, Any] = status or {}

    # -- serialization ---------------------------------------------------
    def to_dict(self) -> dict:
        return {
            "apiVersion": self.api_version,
            "kind": self.kind,
            "metadata": {
                "name": self.metadata.name,
                "namespace": self.metadata.namespace,
                "uid": self.metadata.uid,
                "labels": self.metadata.labels,
                "ownerReferences": self.metadata.owner_references,
                "finalizers": self.metadata.finalizers,
                "creationTimestamp": self.metadata.creation_timestamp,
                "deletionTimestamp": self.metadata.deletion_timestamp,
            },
            "spec": copy.deepcopy(self.spec),
            "status": copy.deepcopy(self.status),
        }

    @classmethod
    def from_dict(cls, d: dict) -> "ApiObject":
        md = d.get("metadata", {})
        meta = ObjectMeta(
            name=md.get("name", ""), namespace=md.get("namespace", "default"),
            uid=md.get("uid", ""), labels=md.get("labels", {}) or {},
            owner_references=md.get("ownerReferences", []) or [],
            finalizers=md.get("finalizers", []) or [],
            creation_timestamp=md.get("creationTimestamp", _now()),
            deletion_timestamp=md.get("deletionTimestamp"))
        return cls(spec=d.get("spec", {}) or {},
                   status=d.get("status", {}) or {}, metadata=meta)

    @property
    def name(self):
        return self.metadata.name

    @property
    def namespace(self):
        return self.metadata.namespace

    def set_owner(self, owner: "ApiObject"):
        self.metadata.owner_references = [{
            "apiVersion": owner.api_version, "kind": owner.kind,
            "name": owner.name, "uid": owner.metadata.uid,
            "controller": True,
        }]


# ------------------------------- finetune.datatunerx.io -----------------
class Finetune(ApiObject):
    """spec: llm, dataset, hyperparameter{hyperparameterRef, overrides},
    image{name,path,imagePullPolicy}, node, resource.
    status: state, trainJobInfo (replaces rayJobInfo), llmCheckpoint{
    llmCheckpointRef, checkpointPath} (finetune_controller.go:115-234)."""
    kind = "Finetune"
    api_version = GROUP_FINETUNE

    STATES = ("Init", "Pending", "Running", "Successful", "Failed")


class FinetuneJob(ApiObject):
    """spec: fineTune{name, finetuneSpec}, scoringPluginConfig{name,
    parameters}, serveConfig. status: state, finetuneStatus, result{
    modelExportResult, image, serve, dashboard, score}, stats
    (finetunejob_controller.go:71-143)."""
    kind = "FinetuneJob"
    api_version = GROUP_FINETUNE

    STATES = ("Init", "Finetune", "BuildImage", "Serve", "Successful",
              "Failed")


class FinetuneExperiment(ApiObject):
    """spec: finetuneJobs[{name, spec}], pending. status: state,
    jobsStatus[], bestVersion{score, image, llm, hyperparameter, dataset},
    stats (finetuneexperiment_controller.go:54-227)."""
    kind = "FinetuneExperiment"
    api_version = GROUP_FINETUNE

    STATES = ("Pending", "Processing", "Success", "Failed")


# ----------------------------------- core.datatunerx.io -----------------
class LLM(ApiObject):
    kind = "LLM"
    api_version = GROUP_CORE


class Hyperparameter(ApiObject):
    """spec.parameters: scheduler, optimizer, int4, int8, loRA_R,
    loRA_Alpha, loRA_Dropout, learningRate, epochs, blockSize, batchSize,
    warmupRatio, weightDecay, gradAccSteps, trainerType, PEFT, FP16
    (finetune_controller.go:483-506)."""
    kind = "Hyperparameter"
    api_version = GROUP_CORE


class LLMCheckpoint(ApiObject):
    """spec: llm{llmRef,spec}, dataset{datasetRef,spec}, hyperparameter{
    hyperparameterRef,spec}, image, checkpoint (path), checkpointImage
    (finetune_controller.go:621-653)."""
    kind = "LLMCheckpoint"
    api_version = GROUP_CORE


# ------------------------------ extension.datatunerx.io -----------------
class Dataset(ApiObject):
    """spec.datasetMetadata.datasetInfo: subsets[].splits.{train,validate,
    test}.file + features[{name: instruction|response, mapTo}]
    (finetune_controller.go:466-478,655-680)."""
    kind = "Dataset"
    api_version = GROUP_EXTENSION


class Scoring(ApiObject):
    """spec: inferenceService, plugin{loadPlugin, name, parameters};
    status.score (generate.go:331-358)."""
    kind = "Scoring"
    api_version = GROUP_EXTENSION


KIND_MAP = {c.kind: c for c in
            (Finetune, FinetuneJob, FinetuneExperiment, LLM, Hyperparameter,
             LLMCheckpoint, Dataset, Scoring)}


def object_from_dict(d: dict) -> ApiObject:
    cls = KIND_MAP.get(d.get("kind", ""))
    if cls is None:
        raise ValueError(f"unknown kind {d.get('kind')!r}")
    return cls.from_dict(d)


# -------- hyperparameter merge (finetune_controller.go:682-758) ---------
def merge_hyperparameters(base: dict, overrides: Optional[dict]) -> dict:
    """Override non-null fields of Hyperparameter.spec.parameters."""
    out = copy.deepcopy(base or {})
    for k, v in (overrides or {}).items():
        if v is not None:
            out[k] = v
    return out
