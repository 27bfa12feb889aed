"""dtx CLI — kubectl/dtx-ctl-style interface to the file-backed API
(INSTALL.md's `dtx-ctl` equivalent, minus Helm: there is no cluster).

  python -m datatunerx_amd.cli apply -f experiment.yaml --state-dir S
  python -m datatunerx_amd.cli get FinetuneExperiment --state-dir S
  python -m datatunerx_amd.cli delete FinetuneJob myjob --state-dir S
  python -m datatunerx_amd.cli manager --state-dir S --n-gpus 8
"""

from __future__ import annotations

import argparse
import json
import sys

import yaml

from .api.store import Store
from .api.types import KIND_MAP


def main(argv=None):
    ap = argparse.ArgumentParser("dtx")
    ap.add_argument("--state-dir", default="./dtx-state")
    sub = ap.add_subparsers(dest="cmd", required=True)

    p_apply = sub.add_parser("apply")
    p_apply.add_argument("-f", "--filename", required=True)

    p_get = sub.add_parser("get")
    p_get.add_argument("kind")
    p_get.add_argument("name", nargs="?")
    p_get.add_argument("-n", "--namespace", default=None)
    p_get.add_argument("-o", "--output", default="table")

    p_del = sub.add_parser("delete")
    p_del.add_argument("kind")
    p_del.add_argument("name")
    p_del.add_argument("-n", "--namespace", default="default")

    p_mgr = sub.add_parser("manager")
    p_mgr.add_argument("--work-dir", default="./dtx-work")
    p_mgr.add_argument("--n-gpus", type=int, default=8)
    p_mgr.add_argument("--storage-path", default="")
    p_mgr.add_argument("--metrics-export-address", default="")
    p_mgr.add_argument("--once", action="store_true",
                       help="reconcile until settled, then exit")

    p_logs = sub.add_parser(
        "logs", help="print a Finetune's trainer step log (jsonl)")
    p_logs.add_argument("name", help="Finetune (or FinetuneJob) name")
    p_logs.add_argument("-n", "--namespace", default="default")
    p_logs.add_argument("--work-dir", default="./dtx-work")
    p_logs.add_argument("--tail", type=int, default=0,
                        help="only the last N lines")

    p_run = sub.add_parser(
        "run", help="apply manifest(s), reconcile to completion, print "
                    "final statuses (dtx-ctl-style one-shot)")
    p_run.add_argument("-f", "--filename", required=True, action="append")
    p_run.add_argument("--work-dir", default="./dtx-work")
    p_run.add_argument("--n-gpus", type=int, default=8)
    p_run.add_argument("--storage-path", default="")
    p_run.add_argument("--timeout", type=float, default=1800.0)

    args = ap.parse_args(argv)
    store = Store(args.state_dir)

    if args.cmd == "apply":
        if args.filename == "-":
            text = sys.stdin.read()
        else:
            with open(args.filename) as fh:
                text = fh.read()
        for obj in _apply(store, text, args.filename):
            print(f"{obj.kind.lower()}/{obj.name} applied")
    elif args.cmd == "get":
        kind = _resolve_kind(args.kind)
        objs = store.list(kind, args.namespace)
        if args.name:
            objs = [o for o in objs if o.name == args.name]
        if args.output in ("yaml", "json"):
            docs = [o.to_dict() for o in objs]
            print(yaml.safe_dump_all(docs) if args.output == "yaml"
                  else json.dumps(docs, indent=2))
        else:
            print(f"{'NAMESPACE':<12} {'NAME':<32} {'STATE':<12} SCORE")
            for o in objs:
                score = (o.status.get("result", {}) or {}).get("score") \
                    or o.status.get("score") or ""
                print(f"{o.namespace:<12} {o.name:<32} "
                      f"{o.status.get('state', ''):<12} {score}")
    elif args.cmd == "delete":
        store.delete(_resolve_kind(args.kind), args.namespace, args.name)
        print(f"{args.kind}/{args.name} deleted")
    elif args.cmd == "logs":
        import os
        # accept either the Finetune name or its parent job's name
        cands = [args.name, f"{args.name}-finetune"]
        for nm in cands:
            log = os.path.join(args.work_dir, args.namespace, nm,
                               "output", "watch", "trainer_log.jsonl")
            if os.path.exists(log):
                with open(log) as f:
                    lines = f.readlines()
                if args.tail:
                    lines = lines[-args.tail:]
                sys.stdout.write("".join(lines))
                return
        raise SystemExit(
            f"no trainer log under {args.work_dir}/{args.namespace}/"
            f"{{{' | '.join(cands)}}}/output/watch/")
    elif args.cmd == "manager":
        from .api.controllers import ManagerConfig
        from .api.manager import Manager
        cfg = ManagerConfig(
            state_dir=args.state_dir, work_dir=args.work_dir,
            n_gpus=args.n_gpus, storage_path=args.storage_path,
            metrics_export_address=args.metrics_export_address)
        mgr = Manager(cfg)
        if args.once:
            ok = mgr.run_until_settled()
            sys.exit(0 if ok else 1)
        mgr.run()
    elif args.cmd == "run":
        from .api.controllers import ManagerConfig
        from .api.manager import Manager
        from .api.types import (FinetuneExperiment, FinetuneJob, Finetune)
        for fn in args.filename:
            if fn == "-":
                text = sys.stdin.read()
            else:
                with open(fn) as fh:
                    text = fh.read()
            for obj in _apply(store, text, fn):
                print(f"{obj.kind.lower()}/{obj.name} applied")
        cfg = ManagerConfig(state_dir=args.state_dir,
                            work_dir=args.work_dir, n_gpus=args.n_gpus,
                            storage_path=args.storage_path)
        mgr = Manager(cfg)
        ok = mgr.run_until_settled(timeout=args.timeout)
        for cls in (FinetuneExperiment, FinetuneJob, Finetune):
            for o in store.list(cls):
                extra = ""
                if cls is FinetuneExperiment and o.status.get("bestVersion"):
                    bv = o.status["bestVersion"]
                    extra = (f"  bestVersion: score={bv.get('score')} "
                             f"image={bv.get('image')}")
                print(f"{o.kind.lower()}/{o.name}: "
                      f"{o.status.get('state', '')}{extra}")
        sys.exit(0 if ok else 1)


def _apply(store, text: str, source: str):
    """apply_manifest with kubectl-style clean errors (bad YAML, unknown
    kind, admission rejection) instead of tracebacks."""
    from .api.validation import ValidationError
    try:
        return store.apply_manifest(text)
    except yaml.YAMLError as e:
        raise SystemExit(f"error: {source}: invalid YAML: {e}")
    except (ValidationError, ValueError) as e:
        raise SystemExit(f"error: {source}: {e}")


def _resolve_kind(k: str) -> str:
    for kind in KIND_MAP:
        if kind.lower() == k.lower():
            return kind
    raise SystemExit(f"unknown kind {k!r}; have {sorted(KIND_MAP)}")


if __name__ == "__main__":
    main()
