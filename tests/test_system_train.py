"""System test of the fine-tune path: Dataset → base Model → fine-tuned
Model (trainer Job with base-model + dataset mounts) — the reference's
finetuned-model.yaml flow (examples/llama2-7b/finetuned-model.yaml) run for
real on CPU, including a 2-process torchrun/gloo variant exercising the
multi-GPU Job wiring."""

import json
import os
import time

import pytest

from substratus_amd.api import types as apiv1
from substratus_amd.cloud import new_cloud
from substratus_amd.controller.manager import Manager
from substratus_amd.kube.memapi import MemKube
from substratus_amd.runtime.local import LocalRuntime
from substratus_amd.sci.base import FakeSCIClient


@pytest.mark.timeout(600)
def test_finetune_flow(tmp_path):
    bucket = str(tmp_path / "bucket")
    os.makedirs(bucket)
    kube = MemKube()
    cloud = new_cloud({"CLOUD": "kind", "CLUSTER_NAME": "tr",
                       "BUCKET_DIR": bucket})
    mgr = Manager(kube, cloud, FakeSCIClient())
    rt = LocalRuntime(kube, bucket)
    kube.create(apiv1.new_object("Dataset", "ds", {
        "image": "substratus-amd/dataset-loader",
        "params": {"synthetic_rows": 64},
    }))
    kube.create(apiv1.new_object("Model", "base", {
        "image": "substratus-amd/model-loader",
        "params": {"name": "llama-tiny"},
    }))
    kube.create(apiv1.new_object("Model", "tuned", {
        "image": "substratus-amd/model-trainer",
        "model": {"name": "base"},
        "dataset": {"name": "ds"},
        "params": {"num_train_epochs": 1, "micro_batch": 2, "seq_len": 32,
                   "steps_per_epoch": 4, "save_steps": 2},
    }))
    deadline = time.time() + 420
    while time.time() < deadline:
        rt.run_until_quiet(mgr, timeout=15)
        if apiv1.get_status_ready(kube.get("Model", "tuned")):
            break
        time.sleep(1)
    try:
        tuned = kube.get("Model", "tuned")
        assert apiv1.get_status_ready(tuned), tuned.get("status")
        url = cloud.object_artifact_url(tuned)
        art = os.path.join(bucket, url.rsplit("/", 1)[-1], "artifacts")
        files = os.listdir(art)
        assert "completed.json" in files
        ckpts = [f for f in files if f.startswith("checkpoint-")]
        assert ckpts, files
        newest = max(ckpts, key=lambda c: int(c.split("-")[1]))
        assert os.path.exists(os.path.join(art, newest, "model.safetensors"))
        assert os.path.exists(os.path.join(art, newest, "trainer_state.json"))
    finally:
        rt.stop()


@pytest.mark.timeout(600)
def test_finetune_multiproc_gloo(tmp_path):
    """gpu.count=2 Job → runtime launches torchrun --nproc-per-node 2; on CPU
    the trainer falls back to gloo DDP — the multi-GPU wiring end-to-end."""
    bucket = str(tmp_path / "bucket")
    os.makedirs(bucket)
    kube = MemKube()
    cloud = new_cloud({"CLOUD": "kind", "CLUSTER_NAME": "tr2",
                       "BUCKET_DIR": bucket})
    mgr = Manager(kube, cloud, FakeSCIClient())
    rt = LocalRuntime(kube, bucket)
    kube.create(apiv1.new_object("Model", "tuned2", {
        "image": "substratus-amd/model-trainer",
        "params": {"name": "llama-tiny", "num_train_epochs": 1,
                   "micro_batch": 1, "seq_len": 16, "steps_per_epoch": 2},
        "resources": {"gpu": {"type": "amd-mi355x", "count": 2}},
    }))
    deadline = time.time() + 420
    while time.time() < deadline:
        rt.run_until_quiet(mgr, timeout=15)
        if apiv1.get_status_ready(kube.get("Model", "tuned2")):
            break
        time.sleep(1)
    try:
        tuned = kube.get("Model", "tuned2")
        assert apiv1.get_status_ready(tuned), tuned.get("status")
        job = kube.get("Job", "tuned2-modeller")
        res = job["spec"]["template"]["spec"]["containers"][0]["resources"]
        assert res["limits"]["amd.com/gpu"] == "2"
    finally:
        rt.stop()
