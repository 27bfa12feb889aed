"""On-hardware control-plane e2e (gpu-marked): the manager gang-allocates
a REAL GPU from the C++ inventory, launches the trainer process (HIP
kernels, bf16), builds the bundle, serves on the GPU and scores —
the full FinetuneJob cascade with cpu_mode OFF."""

import os
import time

import pytest
import torch

pytestmark = pytest.mark.gpu

from datatunerx_amd.api.controllers import ManagerConfig  # noqa: E402
from datatunerx_amd.api.manager import Manager  # noqa: E402
from datatunerx_amd.api.types import (Dataset, Finetune, FinetuneJob,  # noqa: E402
                                      Hyperparameter, LLM)


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs GPU")
def test_full_pipeline_on_gpu(tmp_path):
    cfg = ManagerConfig(state_dir=str(tmp_path / "state"),
                        work_dir=str(tmp_path / "work"),
                        n_gpus=torch.cuda.device_count(),
                        cpu_mode=False, base_port=31300,
                        storage_path=str(tmp_path / "storage"))
    mgr = Manager(cfg)
    mgr.store.create(LLM(name="llama-mini", spec={"family": "llama"}))
    mgr.store.create(Hyperparameter(name="hp", spec={"parameters": {
        "learningRate": "1e-3", "epochs": 1, "blockSize": 128,
        "batchSize": 4, "loRA_R": 8, "loRA_Alpha": 16,
        "loRA_Dropout": "0.0", "maxSteps": 3, "syntheticExamples": 32}}))
    mgr.store.create(Dataset(name="ds", spec={
        "datasetMetadata": {"datasetInfo": {
            "subsets": [{"splits": {"train": {"file": ""}}}],
            "features": []}}}))
    job = FinetuneJob(name="gpujob", spec={
        "fineTune": {"finetuneSpec": {
            "llm": "llama-mini", "dataset": "ds", "node": 1,
            "hyperparameter": {"hyperparameterRef": "hp"}}}})
    mgr.store.create(job)
    deadline = time.time() + 420
    try:
        while time.time() < deadline:
            mgr.reconcile_once()
            cur = mgr.store.get(FinetuneJob, "default", "gpujob")
            if cur.status.get("state") in ("Successful", "Failed"):
                break
            time.sleep(0.5)
        cur = mgr.store.get(FinetuneJob, "default", "gpujob")
        if cur.status.get("state") != "Successful":
            work = os.path.join(str(tmp_path / "work"), "default")
            logs = []
            for root, _, names in os.walk(work):
                for n in names:
                    if n.endswith(".log"):
                        with open(os.path.join(root, n),
                                  errors="replace") as f:
                            logs.append(f"== {n} ==\n" + f.read()[-1500:])
            pytest.fail(f"state={cur.status}\n" + "\n".join(logs))
        assert cur.status["result"].get("score") is not None
        # GPU really allocated + released by the gang inventory
        ft = mgr.store.get(Finetune, "default", "gpujob-finetune")
        assert ft.status["trainJobInfo"]["gpus"] == [0] or \
            len(ft.status["trainJobInfo"]["gpus"]) == 1
    finally:
        # reap anything left (exact pids from statuses)
        for cls in (Finetune, FinetuneJob):
            for obj in mgr.store.list(cls):
                for pid in (obj.status.get("trainJobInfo") or {}).get(
                        "pids", []):
                    try:
                        os.kill(pid, 15)
                    except OSError:
                        pass
                info = obj.status.get("serveInfo") or {}
                for pid in info.get("pids", [info.get("pid")]):
                    if pid:
                        try:
                            os.kill(pid, 15)
                        except OSError:
                            pass
