"""File-backed object store — the apiserver replacement.

Objects live at <root>/<kind>/<namespace>/<name>.yaml. Writes are atomic
(tmp+rename); a monotonically increasing resourceVersion is kept per
object so controllers can detect conflicts (optimistic concurrency, like
controller-runtime's client). Deletion honors finalizers: delete() marks
deletionTimestamp; the object is removed once finalizers are empty.
Cascade delete via ownerReferences is handled by the manager's GC sweep
(manager.py), mirroring Kubernetes GC for the reference's owned objects
(SURVEY.md §3.5).
"""

from __future__ import annotations

import os
import threading
from typing import List, Optional

import yaml

from .types import ApiObject, KIND_MAP, object_from_dict


class Conflict(Exception):
    pass


class NotFound(Exception):
    pass


class Store:
    def __init__(self, root: str):
        self.root = root
        self._lock = threading.RLock()
        os.makedirs(root, exist_ok=True)

    def _path(self, kind: str, namespace: str, name: str) -> str:
        return os.path.join(self.root, kind, namespace, f"{name}.yaml")

    # ------------------------------------------------------------- CRUD
    def create(self, obj: ApiObject) -> ApiObject:
        # admission: defaulting then validating webhooks (native parity
        # with controller_manager.go:112-135)
        from .validation import default_, validate_
        default_(obj)
        validate_(obj)
        with self._lock:
            p = self._path(obj.kind, obj.namespace, obj.name)
            if os.path.exists(p):
                raise Conflict(f"{obj.kind}/{obj.name} exists")
            d = obj.to_dict()
            d["metadata"]["resourceVersion"] = 1
            self._write(p, d)
            return obj

    def get(self, kind_or_cls, namespace: str, name: str) -> ApiObject:
        kind = kind_or_cls if isinstance(kind_or_cls, str) else \
            kind_or_cls.kind
        p = self._path(kind, namespace, name)
        with self._lock:
            if not os.path.exists(p):
                raise NotFound(f"{kind}/{namespace}/{name}")
            with open(p) as f:
                d = yaml.safe_load(f)
        obj = object_from_dict(d)
        obj._rv = d["metadata"].get("resourceVersion", 0)
        return obj

    def try_get(self, kind_or_cls, namespace: str, name: str):
        try:
            return self.get(kind_or_cls, namespace, name)
        except NotFound:
            return None

    def update(self, obj: ApiObject) -> ApiObject:
        """Whole-object update with optimistic concurrency."""
        with self._lock:
            p = self._path(obj.kind, obj.namespace, obj.name)
            if not os.path.exists(p):
                raise NotFound(f"{obj.kind}/{obj.name}")
            with open(p) as f:
                cur = yaml.safe_load(f)
            cur_rv = cur["metadata"].get("resourceVersion", 0)
            if getattr(obj, "_rv", cur_rv) != cur_rv:
                raise Conflict(f"{obj.kind}/{obj.name} rv mismatch")
            d = obj.to_dict()
            d["metadata"]["resourceVersion"] = cur_rv + 1
            self._write(p, d)
            obj._rv = cur_rv + 1
            return obj

    def list(self, kind_or_cls, namespace: Optional[str] = None
             ) -> List[ApiObject]:
        kind = kind_or_cls if isinstance(kind_or_cls, str) else \
            kind_or_cls.kind
        out = []
        base = os.path.join(self.root, kind)
        with self._lock:
            if not os.path.isdir(base):
                return []
            for ns in sorted(os.listdir(base)):
                if namespace and ns != namespace:
                    continue
                nsdir = os.path.join(base, ns)
                for fn in sorted(os.listdir(nsdir)):
                    if fn.endswith(".yaml"):
                        with open(os.path.join(nsdir, fn)) as f:
                            d = yaml.safe_load(f)
                        obj = object_from_dict(d)
                        obj._rv = d["metadata"].get("resourceVersion", 0)
                        out.append(obj)
        return out

    def delete(self, kind_or_cls, namespace: str, name: str):
        """Mark for deletion; remove immediately if no finalizers."""
        import time
        with self._lock:
            obj = self.try_get(kind_or_cls, namespace, name)
            if obj is None:
                return
            if obj.metadata.finalizers:
                if obj.metadata.deletion_timestamp is None:
                    obj.metadata.deletion_timestamp = time.strftime(
                        "%Y-%m-%dT%H:%M:%SZ", time.gmtime())
                    self.update(obj)
            else:
                os.remove(self._path(obj.kind, namespace, name))

    def remove_now(self, kind_or_cls, namespace: str, name: str):
        kind = kind_or_cls if isinstance(kind_or_cls, str) else \
            kind_or_cls.kind
        with self._lock:
            p = self._path(kind, namespace, name)
            if os.path.exists(p):
                os.remove(p)

    # ------------------------------------------------------------ utils
    def _write(self, path: str, d: dict):
        os.makedirs(os.path.dirname(path), exist_ok=True)
        tmp = path + ".tmp"
        with open(tmp, "w") as f:
            yaml.safe_dump(d, f, sort_keys=False)
        os.replace(tmp, path)

    def apply_manifest(self, text: str) -> List[ApiObject]:
        """kubectl-apply-style: create or update from YAML doc(s)."""
        out = []
        for doc in yaml.safe_load_all(text):
            if not doc:
                continue
            obj = object_from_dict(doc)
            existing = self.try_get(obj.kind, obj.namespace, obj.name)
            if existing is None:
                out.append(self.create(obj))
            else:
                existing.spec = obj.spec
                out.append(self.update(existing))
        return out

    def gc_sweep(self):
        """Remove objects whose controller owner no longer exists
        (Kubernetes-GC equivalent for ownerReferences)."""
        for kind in KIND_MAP:
            for obj in self.list(kind):
                for ref in obj.metadata.owner_references:
                    owner = self.try_get(ref["kind"], obj.namespace,
                                         ref["name"])
                    if owner is None or (owner.metadata.uid and ref.get("uid")
                                         and owner.metadata.uid != ref["uid"]):
                        self.delete(kind, obj.namespace, obj.name)
                        break
