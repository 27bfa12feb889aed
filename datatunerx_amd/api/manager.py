"""Controller manager: the native replacement for the reference's
cmd/controller-manager (controller_manager.go:53-175). Runs the four
reconcilers level-triggered over the file store, with per-object requeue
timing (handlererr parity) and an ownerReference GC sweep.
"""

from __future__ import annotations

import argparse
import threading
import time
from typing import Dict

from ..native import _dtx_native
from .controllers import (FinetuneController, FinetuneExperimentController,
                          FinetuneJobController, ManagerConfig,
                          PortAllocator, REQUEUE_ERROR, ScoringController)
from .store import Conflict, Store
from .types import (Finetune, FinetuneExperiment, FinetuneJob, Scoring)


class Manager:
    def __init__(self, cfg: ManagerConfig):
        self.cfg = cfg
        self.store = Store(cfg.state_dir)
        self.inventory = _dtx_native.GpuInventory(cfg.n_gpus)
        self.supervisor = _dtx_native.ProcessSupervisor()
        ports = PortAllocator(cfg.base_port)
        self.controllers = [
            (FinetuneExperiment,
             FinetuneExperimentController(self.store)),
            (FinetuneJob,
             FinetuneJobController(self.store, self.inventory,
                                   self.supervisor, cfg, ports)),
            (Finetune,
             FinetuneController(self.store, self.inventory,
                                self.supervisor, cfg, ports)),
            (Scoring, ScoringController(self.store)),
        ]
        self._not_before: Dict[str, float] = {}
        self.stop_event = threading.Event()
        self.counters: Dict[str, int] = {"reconcile_total": 0,
                                         "reconcile_errors_total": 0}

    def reconcile_once(self) -> int:
        """One pass over every object; returns number reconciled."""
        n = 0
        now = time.time()
        for cls, ctrl in self.controllers:
            for obj in self.store.list(cls):
                key = f"{cls.kind}/{obj.namespace}/{obj.name}"
                if self._not_before.get(key, 0) > now:
                    continue
                try:
                    requeue = ctrl.reconcile(obj)
                except Conflict:
                    requeue = 1
                except Exception:
                    import traceback
                    traceback.print_exc()
                    requeue = REQUEUE_ERROR
                    self.counters["reconcile_errors_total"] += 1
                n += 1
                self.counters["reconcile_total"] += 1
                if requeue:
                    self._not_before[key] = now + requeue
                else:
                    self._not_before.pop(key, None)
        self.store.gc_sweep()
        return n

    def run(self, poll_interval: float = 1.0):
        while not self.stop_event.is_set():
            self.reconcile_once()
            self.stop_event.wait(poll_interval)

    def serve_metrics(self, host: str = "127.0.0.1", port: int = 8080):
        """Prometheus text endpoint (controller-runtime's :8080 metrics
        parity — options.go:12) + /healthz, /readyz. Runs in a daemon
        thread; returns the server."""
        from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
        mgr = self

        class H(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def do_GET(self):
                if self.path in ("/healthz", "/readyz"):
                    body = b"ok"
                elif self.path == "/metrics":
                    lines = []
                    for k, v in mgr.counters.items():
                        lines.append(f"# TYPE dtx_{k} counter")
                        lines.append(f"dtx_{k} {v}")
                    for cls, _ in mgr.controllers:
                        objs = mgr.store.list(cls)
                        states: Dict[str, int] = {}
                        for o in objs:
                            st = o.status.get("state", "") or "none"
                            states[st] = states.get(st, 0) + 1
                        for st, c in states.items():
                            lines.append(
                                f'dtx_objects{{kind="{cls.kind}",'
                                f'state="{st}"}} {c}')
                    body = ("\n".join(lines) + "\n").encode()
                else:
                    self.send_response(404)
                    self.end_headers()
                    return
                self.send_response(200)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

        srv = ThreadingHTTPServer((host, port), H)
        t = threading.Thread(target=srv.serve_forever, daemon=True)
        t.start()
        return srv

    def run_until_settled(self, timeout: float = 300.0,
                          poll_interval: float = 0.2) -> bool:
        """Drive reconciliation until no object is in a non-terminal
        state (used by tests and one-shot CLI runs)."""
        deadline = time.time() + timeout
        terminal = {"Successful", "Failed", "Success", "Pending"}
        while time.time() < deadline:
            self.reconcile_once()
            busy = False
            for cls, _ in self.controllers:
                for obj in self.store.list(cls):
                    st = obj.status.get("state", "")
                    if cls is Scoring:
                        if obj.status.get("score") is None:
                            busy = True
                    elif st not in terminal:
                        busy = True
            if not busy:
                return True
            time.sleep(poll_interval)
        return False


def main(argv=None):
    ap = argparse.ArgumentParser("datatunerx_amd controller manager")
    ap.add_argument("--state-dir", default="./dtx-state")
    ap.add_argument("--work-dir", default="./dtx-work")
    ap.add_argument("--n-gpus", type=int, default=8)
    ap.add_argument("--storage-path", default="")
    ap.add_argument("--metrics-export-address", default="")
    ap.add_argument("--poll-interval", type=float, default=1.0)
    ap.add_argument("--metrics-port", type=int, default=8080,
                    help="0 disables the /metrics endpoint")
    args = ap.parse_args(argv)
    cfg = ManagerConfig(state_dir=args.state_dir, work_dir=args.work_dir,
                        n_gpus=args.n_gpus, storage_path=args.storage_path,
                        metrics_export_address=args.metrics_export_address)
    mgr = Manager(cfg)
    if args.metrics_port:
        mgr.serve_metrics(port=args.metrics_port)
    mgr.run(args.poll_interval)


if __name__ == "__main__":
    main()
