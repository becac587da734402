"""System test — the reference's test/system.sh end-to-end on CPU with no
cluster: apply a base Model + Server, drive the controllers + local runtime
until .status.ready, then POST /v1/completions and get text back
(reference test/system.sh:28-77; facebook-opt-125m stands in as llama-tiny
for CPU speed — the reference used opt-125m for the same reason)."""

import json
import os
import time

import httpx
import pytest

from substratus_amd.api import types as apiv1
from substratus_amd.cloud import new_cloud
from substratus_amd.controller.manager import Manager
from substratus_amd.kube.memapi import MemKube
from substratus_amd.runtime.local import LocalRuntime
from substratus_amd.sci.base import FakeSCIClient


@pytest.mark.timeout(300)
def test_system_model_server_completion(tmp_path):
    bucket = str(tmp_path / "bucket")
    os.makedirs(bucket)
    kube = MemKube()
    cloud = new_cloud({"CLOUD": "kind", "CLUSTER_NAME": "sys",
                       "BUCKET_DIR": bucket})
    mgr = Manager(kube, cloud, FakeSCIClient())
    rt = LocalRuntime(kube, bucket)

    # kubectl apply -f base-model.yaml (reference examples/*/base-model.yaml)
    kube.create(apiv1.new_object("Model", "model-tiny", {
        "image": "substratus-amd/model-loader",
        "params": {"name": "llama-tiny"},
    }))
    # kubectl apply -f server.yaml
    kube.create(apiv1.new_object("Server", "server-tiny", {
        "image": "substratus-amd/model-server",
        "model": {"name": "model-tiny"},
        "params": {"max_length": 128, "max_batch": 2},
    }))

    # kubectl wait .status.ready=true (720s budget in reference)
    deadline = time.time() + 240
    while time.time() < deadline:
        rt.run_until_quiet(mgr, timeout=10)
        model = kube.get("Model", "model-tiny")
        server = kube.get("Server", "server-tiny")
        if apiv1.get_status_ready(model) and apiv1.get_status_ready(server):
            break
        time.sleep(1)
    try:
        assert apiv1.get_status_ready(kube.get("Model", "model-tiny")), \
            kube.get("Model", "model-tiny").get("status")
        assert apiv1.get_status_ready(kube.get("Server", "server-tiny")), \
            kube.get("Server", "server-tiny").get("status")

        # loader wrote artifacts into the deterministic bucket path
        url = cloud.object_artifact_url(kube.get("Model", "model-tiny"))
        digest = url.rsplit("/", 1)[-1]
        art = os.path.join(bucket, digest, "artifacts")
        assert os.path.exists(os.path.join(art, "model.safetensors"))
        assert os.path.exists(os.path.join(art, "completed.json"))

        # the system test's completion request (test/system.sh:73-77)
        svc = kube.get("Service", "server-tiny-server")
        endpoint = svc["metadata"]["annotations"][
            "substratus.ai/local-endpoint"]
        r = httpx.post(f"{endpoint}/v1/completions", json={
            "prompt": "Who was the first president of the United States? ",
            "max_tokens": 8}, timeout=120)
        assert r.status_code == 200
        body = r.json()
        assert body["usage"]["completion_tokens"] >= 1
    finally:
        rt.stop()


@pytest.mark.timeout(180)
def test_notebook_flow(tmp_path):
    """Notebook CR → local runtime runs the notebook shim → readiness GET
    /api → Ready; suspend kills it (reference notebook lifecycle)."""
    bucket = str(tmp_path / "bucket")
    os.makedirs(bucket)
    kube = MemKube()
    cloud = new_cloud({"CLOUD": "kind", "CLUSTER_NAME": "nb",
                       "BUCKET_DIR": bucket})
    mgr = Manager(kube, cloud, FakeSCIClient())
    rt = LocalRuntime(kube, bucket)
    kube.create(apiv1.new_object("Notebook", "nb1", {
        "image": "substratus-amd/base",
    }))
    deadline = time.time() + 120
    ready = False
    try:
        while time.time() < deadline:
            rt.run_until_quiet(mgr, timeout=5)
            nb = kube.get("Notebook", "nb1")
            if apiv1.get_status_ready(nb):
                ready = True
                break
            time.sleep(0.5)
        assert ready, kube.get("Notebook", "nb1").get("status")
        pod = kube.get("Pod", "nb1-notebook")
        ep = pod["metadata"]["annotations"]["substratus.ai/local-endpoint"]
        r = httpx.get(f"{ep}/api", timeout=5)
        assert r.status_code == 200
        # file write/read through the shim (NOTEBOOK_TOKEN honored like
        # jupyter's ?token= — the controller sets it to "default")
        httpx.put(f"{ep}/files/hello.py?token=default",
                  content=b"print(1)\n", timeout=5).raise_for_status()
        assert "hello.py" in httpx.get(f"{ep}/files?token=default",
                                       timeout=5).json()["files"]
        # wrong token -> 403 (Jupyter ?token= behavior)
        assert httpx.get(f"{ep}/files?token=WRONG",
                         timeout=5).status_code == 403
        # suspend deletes the pod AND the runtime reaps the shim process
        kube.patch("Notebook", "nb1", {"spec": {"suspend": True}})
        mgr.run_until_settled()
        assert kube.try_get("Pod", "nb1-notebook") is None
        rt.sync()
        assert ("Pod", "default", "nb1-notebook") not in rt.procs
        try:
            httpx.get(f"{ep}/api", timeout=2)
            alive = True
        except Exception:
            alive = False
        assert not alive, "notebook shim should be terminated on suspend"
    finally:
        rt.stop()


@pytest.mark.timeout(600)
def test_system_opt125m_like_reference(tmp_path):
    """BASELINE config 1 with the reference's actual model: facebook-opt-125m
    Model+Server on CPU, completion request (test/system.sh's exact shape)."""
    bucket = str(tmp_path / "bucket")
    os.makedirs(bucket)
    kube = MemKube()
    cloud = new_cloud({"CLOUD": "kind", "CLUSTER_NAME": "sys2",
                       "BUCKET_DIR": bucket})
    mgr = Manager(kube, cloud, FakeSCIClient())
    rt = LocalRuntime(kube, bucket)
    kube.create(apiv1.new_object("Model", "facebook-opt-125m", {
        "image": "substratus-amd/model-loader",
        "params": {"name": "opt-125m", "dtype": "fp32"},
    }))
    kube.create(apiv1.new_object("Server", "facebook-opt-125m", {
        "image": "substratus-amd/model-server",
        "model": {"name": "facebook-opt-125m"},
        "params": {"max_length": 64, "max_batch": 2},
    }))
    deadline = time.time() + 480
    try:
        while time.time() < deadline:
            rt.run_until_quiet(mgr, timeout=15)
            if (apiv1.get_status_ready(kube.get("Model", "facebook-opt-125m"))
                    and apiv1.get_status_ready(
                        kube.get("Server", "facebook-opt-125m"))):
                break
            time.sleep(1)
        assert apiv1.get_status_ready(kube.get("Server", "facebook-opt-125m"))
        svc = kube.get("Service", "facebook-opt-125m-server")
        ep = svc["metadata"]["annotations"]["substratus.ai/local-endpoint"]
        r = httpx.post(f"{ep}/v1/completions", json={
            "prompt": "Who was the first president of the United States?",
            "max_tokens": 4}, timeout=180)
        assert r.status_code == 200
        assert r.json()["usage"]["completion_tokens"] >= 1
    finally:
        rt.stop()


@pytest.mark.timeout(180)
def test_hbm_admission_end_to_end(tmp_path):
    """Full reconcile+runtime run, no status hand-patching (VERDICT r01
    next-round #2): the loader's completed.json carries sizeBytes, the Model
    controller surfaces it, and a 600 GiB model on gpu.count=1 yields
    InsufficientHBM while count=8 admits. The huge model is simulated with
    resume semantics: a pre-existing completed.json in the bucket (the
    loader Job skips, exactly like a re-created Model resuming from its
    deterministic artifact path)."""
    bucket = str(tmp_path / "bucket")
    os.makedirs(bucket)
    kube = MemKube()
    cloud = new_cloud({"CLOUD": "kind", "CLUSTER_NAME": "sys",
                       "BUCKET_DIR": bucket})
    mgr = Manager(kube, cloud, FakeSCIClient())
    rt = LocalRuntime(kube, bucket)

    big = apiv1.new_object("Model", "model-big", {
        "image": "substratus-amd/model-loader",
        "params": {"name": "llama-tiny"},
    })
    # pre-seed the deterministic artifact path: 600 GiB already "loaded"
    url = cloud.object_artifact_url(big)
    art = os.path.join(bucket, url.rsplit("/", 1)[-1], "artifacts")
    os.makedirs(art)
    with open(os.path.join(art, "completed.json"), "w") as f:
        json.dump({"completed": True, "name": "llama2-70b",
                   "sizeBytes": 600 * (1 << 30)}, f)
    kube.create(big)
    kube.create(apiv1.new_object("Server", "server-big", {
        "image": "substratus-amd/model-server",
        "model": {"name": "model-big"},
        "resources": {"gpu": {"type": "amd-mi355x", "count": 1}},
    }))
    try:
        deadline = time.time() + 120
        while time.time() < deadline:
            rt.run_until_quiet(mgr, timeout=10)
            if apiv1.get_status_ready(kube.get("Model", "model-big")):
                break
            time.sleep(0.5)
        model = kube.get("Model", "model-big")
        assert apiv1.get_status_ready(model)
        assert model["status"]["artifacts"]["sizeBytes"] == 600 * (1 << 30)
        mgr.run_until_settled()
        from substratus_amd.api import conditions as cond
        s = kube.get("Server", "server-big")
        c = apiv1.get_condition(s, cond.DEPLOYED)
        assert c["reason"] == "InsufficientHBM", c
        assert kube.try_get("Deployment", "server-big-server") is None
        # 8 GPUs (8 x 288 GB) admit the same model
        kube.patch("Server", "server-big", {"spec": {"resources": {
            "gpu": {"type": "amd-mi355x", "count": 8}}}})
        mgr.run_until_settled()
        assert kube.try_get("Deployment", "server-big-server") is not None
    finally:
        rt.stop()


@pytest.mark.timeout(180)
def test_loader_reports_real_size(tmp_path):
    """Positive path of the sizeBytes chain: the real loader writes actual
    artifact bytes and the Model status carries them."""
    bucket = str(tmp_path / "bucket")
    os.makedirs(bucket)
    kube = MemKube()
    cloud = new_cloud({"CLOUD": "kind", "CLUSTER_NAME": "sys",
                       "BUCKET_DIR": bucket})
    mgr = Manager(kube, cloud, FakeSCIClient())
    rt = LocalRuntime(kube, bucket)
    kube.create(apiv1.new_object("Model", "model-size", {
        "image": "substratus-amd/model-loader",
        "params": {"name": "llama-tiny"},
    }))
    try:
        deadline = time.time() + 120
        while time.time() < deadline:
            rt.run_until_quiet(mgr, timeout=10)
            if apiv1.get_status_ready(kube.get("Model", "model-size")):
                break
            time.sleep(0.5)
        model = kube.get("Model", "model-size")
        assert apiv1.get_status_ready(model)
        url = cloud.object_artifact_url(model)
        art = os.path.join(bucket, url.rsplit("/", 1)[-1], "artifacts")
        expect = sum(os.path.getsize(os.path.join(art, f))
                     for f in os.listdir(art) if f != "completed.json")
        assert expect > 0
        assert model["status"]["artifacts"]["sizeBytes"] == expect
    finally:
        rt.stop()
