"""Control-plane tests mirroring the reference's test strategy:
unit URL-scheme/env tests (reference internal/cloud/common_test.go,
internal/controller/utils_test.go) + integration tests against the in-memory
API with faked Job/Pod/Deployment statuses (reference
internal/controller/main_test.go:245-265 fakeJobComplete / fakePodReady)."""

import hashlib

import pytest

from substratus_amd.api import conditions as cond
from substratus_amd.api import types as apiv1
from substratus_amd.cloud import new_cloud
from substratus_amd.cloud.resources import (ResourceError, apply_resources,
                                            check_hbm_fit)
from substratus_amd.controller.manager import Manager
from substratus_amd.controller.utils import param_env, resolve_env
from substratus_amd.kube.memapi import MemKube
from substratus_amd.sci.base import FakeSCIClient


KIND_ENV = {"CLOUD": "kind", "CLUSTER_NAME": "test", "BUCKET_DIR": "/bucket"}


@pytest.fixture()
def harness():
    kube = MemKube()
    cloud = new_cloud(KIND_ENV)
    sci = FakeSCIClient()
    mgr = Manager(kube, cloud, sci)
    return kube, cloud, sci, mgr


def fake_job_complete(kube, name, ns="default"):
    kube.patch("Job", name, {"status": {"conditions": [
        {"type": "Complete", "status": "True"}]}}, ns)


def fake_job_failed(kube, name, ns="default"):
    kube.patch("Job", name, {"status": {"conditions": [
        {"type": "Failed", "status": "True"}]}}, ns)


def fake_deployment_ready(kube, name, ns="default"):
    kube.patch("Deployment", name, {"status": {"readyReplicas": 1}}, ns)


def fake_pod_ready(kube, name, ns="default"):
    kube.patch("Pod", name, {"status": {"conditions": [
        {"type": "Ready", "status": "True"}]}}, ns)


# ------------------------------------------------------------ URL schemes
def test_image_url_scheme():
    cloud = new_cloud(KIND_ENV)
    obj = apiv1.new_object("Model", "m1")
    obj["spec"]["build"] = {"git": {"url": "https://x", "tag": "v1"}}
    assert cloud.object_built_image_url(obj) == \
        "registry:5000/test-model-default-m1:v1"
    obj["spec"]["build"] = {"git": {"url": "https://x", "branch": "dev"}}
    assert cloud.object_built_image_url(obj).endswith(":dev")
    obj["spec"]["build"] = {"upload": {"md5checksum": "abc123"}}
    assert cloud.object_built_image_url(obj).endswith(":abc123")
    obj["spec"]["build"] = {}
    assert cloud.object_built_image_url(obj).endswith(":latest")


def test_artifact_url_deterministic_hash():
    """Exact-hash assertion like reference internal/cloud/common_test.go:16-77:
    the artifact URL is keyed by md5 of the name path so delete/recreate
    resumes from the bucket."""
    cloud = new_cloud(KIND_ENV)
    obj = apiv1.new_object("Model", "m1")
    expect = hashlib.md5(
        b"clusters/test/namespaces/default/models/m1").hexdigest()
    assert cloud.object_artifact_url(obj) == f"tar:///bucket/{expect}"
    # same name -> same URL after recreate
    assert cloud.object_artifact_url(apiv1.new_object("Model", "m1")) == \
        cloud.object_artifact_url(obj)


# ----------------------------------------------------------------- env/params
def test_resolve_env_secret_refs():
    """reference internal/controller/utils_test.go:20-66 table."""
    env = {"PLAIN": "v", "SECRET": "${{ secrets.mysecret.mykey }}"}
    out = resolve_env(env)
    assert {"name": "PLAIN", "value": "v"} in out
    assert {"name": "SECRET", "valueFrom": {"secretKeyRef": {
        "name": "mysecret", "key": "mykey"}}} in out


def test_param_env_uppercases():
    out = param_env({"name": "llama2-7b", "max-length": 100})
    assert {"name": "PARAM_NAME", "value": "llama2-7b"} in out
    assert {"name": "PARAM_MAX_LENGTH", "value": "100"} in out


# ----------------------------------------------------------------- resources
def test_apply_resources_amd_gpu():
    spec = {"containers": [{"name": "x"}]}
    apply_resources(spec, {"cpu": 4, "memory": 48, "disk": 100,
                           "gpu": {"type": "amd-mi355x", "count": 8}}, "kind")
    res = spec["containers"][0]["resources"]
    assert res["limits"]["amd.com/gpu"] == "8"
    assert res["requests"]["memory"] == "48Gi"


def test_apply_resources_rejects_multinode():
    spec = {"containers": [{"name": "x"}]}
    with pytest.raises(ResourceError):
        apply_resources(spec, {"gpu": {"type": "amd-mi355x", "count": 16}},
                        "kind")


def test_hbm_admission():
    check_hbm_fit("amd-mi355x", 1, 100 << 30)        # 100 GiB fits 288
    with pytest.raises(ResourceError):
        check_hbm_fit("amd-mi355x", 1, 300 << 30)    # 300 GiB doesn't
    check_hbm_fit("amd-mi355x", 8, 2000 << 30)       # TP=8: 2 TB fits 2304


# --------------------------------------------------------------- validation
def test_validation():
    obj = apiv1.new_object("Server", "s1", {"image": "x"})
    assert any("model.name" in e for e in apiv1.validate(obj))
    obj["spec"]["model"] = {"name": "m1"}
    assert apiv1.validate(obj) == []
    bad = apiv1.new_object("Model", "m1", {"build": {
        "git": {"url": "u"}, "upload": {"md5checksum": "x", "requestID": "r"}}})
    assert any("exactly one" in e for e in apiv1.validate(bad))


# --------------------------------------------------------- model reconcile
def test_model_direct_image_to_ready(harness):
    kube, cloud, sci, mgr = harness
    kube.create(apiv1.new_object("Model", "m1", {
        "image": "substratus/loader:v1",
        "params": {"name": "llama2-7b"},
    }))
    mgr.run_until_settled()
    job = kube.get("Job", "m1-modeller")
    env = {e["name"]: e.get("value") for e in
           job["spec"]["template"]["spec"]["containers"][0]["env"]}
    assert env["PARAM_NAME"] == "llama2-7b"
    assert job["spec"]["backoffLimit"] == 2  # cpu<=3, no gpu -> retried
    # params.json mounted via subPath
    mounts = job["spec"]["template"]["spec"]["containers"][0]["volumeMounts"]
    assert any(m["mountPath"] == "/content/params.json" for m in mounts)
    assert job["spec"]["template"]["spec"]["securityContext"]["fsGroup"] == 3003
    m = kube.get("Model", "m1")
    assert not apiv1.get_status_ready(m)
    fake_job_complete(kube, "m1-modeller")
    mgr.run_until_settled()
    m = kube.get("Model", "m1")
    assert apiv1.get_status_ready(m)
    assert m["status"]["artifacts"]["url"] == cloud.object_artifact_url(m)


def test_model_gpu_job_not_retried(harness):
    kube, cloud, sci, mgr = harness
    kube.create(apiv1.new_object("Model", "m2", {
        "image": "x", "resources": {"gpu": {"type": "amd-mi355x", "count": 4}},
    }))
    mgr.run_until_settled()
    job = kube.get("Job", "m2-modeller")
    assert job["spec"]["backoffLimit"] == 0  # GPU jobs are not retried
    res = job["spec"]["template"]["spec"]["containers"][0]["resources"]
    assert res["limits"]["amd.com/gpu"] == "4"


def test_model_gates_on_dataset_then_trains(harness):
    kube, cloud, sci, mgr = harness
    kube.create(apiv1.new_object("Model", "base", {"image": "x"}))
    kube.create(apiv1.new_object("Dataset", "d1", {"image": "x"}))
    kube.create(apiv1.new_object("Model", "tuned", {
        "image": "trainer:v1", "model": {"name": "base"},
        "dataset": {"name": "d1"},
    }))
    mgr.run_until_settled()
    assert kube.try_get("Job", "tuned-modeller") is None  # gated
    fake_job_complete(kube, "base-modeller")
    fake_job_complete(kube, "d1-data-loader")
    mgr.run_until_settled()
    job = kube.get("Job", "tuned-modeller")  # dependency-driven requeue
    mounts = job["spec"]["template"]["spec"]["containers"][0]["volumeMounts"]
    paths = {m["mountPath"]: m.get("readOnly", False) for m in mounts}
    assert paths["/content/model"] is True
    assert paths["/content/data"] is True
    assert paths["/content/artifacts"] is False
    fake_job_complete(kube, "tuned-modeller")
    mgr.run_until_settled()
    assert apiv1.get_status_ready(kube.get("Model", "tuned"))


def test_model_job_failed_condition(harness):
    kube, cloud, sci, mgr = harness
    kube.create(apiv1.new_object("Model", "m3", {"image": "x"}))
    mgr.run_until_settled()
    fake_job_failed(kube, "m3-modeller")
    mgr.run_until_settled()
    m = kube.get("Model", "m3")
    c = apiv1.get_condition(m, cond.MODELLED)
    assert c["status"] == "False" and c["reason"] == cond.REASON_JOB_FAILED


# --------------------------------------------------------- build reconcile
def test_build_upload_flow(harness):
    kube, cloud, sci, mgr = harness
    kube.create(apiv1.new_object("Model", "up1", {
        "build": {"upload": {"md5checksum": "deadbeef", "requestID": "r-1"}},
    }))
    mgr.run_until_settled()
    m = kube.get("Model", "up1")
    up = apiv1.get_status_upload(m)
    assert up["requestID"] == "r-1"
    assert up["signedURL"] == sci.signed_url_value
    assert sci.signed_urls, "SCI CreateSignedURL was called"
    assert not apiv1.is_condition_true(m, cond.UPLOADED)
    # client PUTs the tarball; SCI now reports the matching MD5
    bucket, objpath, _ = sci.signed_urls[0]
    sci.md5s[(bucket, objpath)] = "deadbeef"
    kube.patch("Model", "up1", {"metadata": {"annotations": {
        "substratus.ai/upload-timestamp": "now"}}})
    mgr.run_until_settled()
    m = kube.get("Model", "up1")
    assert apiv1.is_condition_true(m, cond.UPLOADED)
    job = kube.get("Job", "up1-model-bld")
    assert "kaniko" in job["spec"]["template"]["spec"]["containers"][0]["image"]
    fake_job_complete(kube, "up1-model-bld")
    mgr.run_until_settled()
    m = kube.get("Model", "up1")
    assert apiv1.get_image(m).endswith(":deadbeef")  # image stamped
    assert apiv1.is_condition_true(m, cond.BUILT)


def test_build_git_flow(harness):
    kube, cloud, sci, mgr = harness
    kube.create(apiv1.new_object("Server", "g1", {
        "model": {"name": "mm"},
        "build": {"git": {"url": "https://github.com/x/y", "branch": "main"}},
    }))
    mgr.run_until_settled()
    job = kube.get("Job", "g1-server-bld")
    init = job["spec"]["template"]["spec"]["initContainers"][0]
    assert "clone" in init["args"][0]
    assert job["spec"]["template"]["spec"]["securityContext"]["runAsUser"] == 0
    fake_job_complete(kube, "g1-server-bld")
    mgr.run_until_settled()
    s = kube.get("Server", "g1")
    assert apiv1.get_image(s).endswith(":main")


# --------------------------------------------------------- server reconcile
def test_server_full_flow(harness):
    kube, cloud, sci, mgr = harness
    kube.create(apiv1.new_object("Model", "m1", {"image": "x"}))
    kube.create(apiv1.new_object("Server", "s1", {
        "image": "server:v1", "model": {"name": "m1"},
    }))
    mgr.run_until_settled()
    assert kube.try_get("Deployment", "s1-server") is None  # gated on model
    fake_job_complete(kube, "m1-modeller")
    mgr.run_until_settled()
    dep = kube.get("Deployment", "s1-server")
    assert dep["spec"]["replicas"] == 1
    c = dep["spec"]["template"]["spec"]["containers"][0]
    assert c["readinessProbe"]["httpGet"]["path"] == "/"
    assert c["ports"][0]["containerPort"] == 8080
    svc = kube.get("Service", "s1-server")
    assert svc["spec"]["ports"][0]["port"] == 8080
    mounts = {m["mountPath"]: m.get("readOnly", False)
              for m in c["volumeMounts"]}
    assert mounts["/content/model"] is True
    s = kube.get("Server", "s1")
    assert not apiv1.get_status_ready(s)
    fake_deployment_ready(kube, "s1-server")
    mgr.run_until_settled()
    assert apiv1.get_status_ready(kube.get("Server", "s1"))


def test_server_hbm_admission_rejects(harness):
    kube, cloud, sci, mgr = harness
    kube.create(apiv1.new_object("Model", "big", {"image": "x"}))
    mgr.run_until_settled()
    fake_job_complete(kube, "big-modeller")
    mgr.run_until_settled()
    # model artifacts report 600 GiB of weights; 1 GPU has 288 GB
    kube.patch("Model", "big", {"status": {"artifacts": {
        "sizeBytes": 600 * (1 << 30)}}})
    kube.create(apiv1.new_object("Server", "sbig", {
        "image": "server:v1", "model": {"name": "big"},
        "resources": {"gpu": {"type": "amd-mi355x", "count": 1}},
    }))
    mgr.run_until_settled()
    s = kube.get("Server", "sbig")
    c = apiv1.get_condition(s, cond.DEPLOYED)
    assert c["reason"] == "InsufficientHBM"
    assert kube.try_get("Deployment", "sbig-server") is None
    # TP=8 fits: 8 x 288 GB
    kube.patch("Server", "sbig", {"spec": {"resources": {"gpu": {
        "type": "amd-mi355x", "count": 8}}}})
    mgr.run_until_settled()
    assert kube.try_get("Deployment", "sbig-server") is not None


# ------------------------------------------------------- dataset reconcile
def test_dataset_flow(harness):
    kube, cloud, sci, mgr = harness
    kube.create(apiv1.new_object("Dataset", "d1", {
        "image": "loader:v1", "params": {"urls": "file:///x.jsonl"},
    }))
    mgr.run_until_settled()
    job = kube.get("Job", "d1-data-loader")
    assert job["spec"]["backoffLimit"] == 2
    fake_job_complete(kube, "d1-data-loader")
    mgr.run_until_settled()
    assert apiv1.get_status_ready(kube.get("Dataset", "d1"))


# ------------------------------------------------------ notebook reconcile
def test_notebook_suspend_resume(harness):
    kube, cloud, sci, mgr = harness
    kube.create(apiv1.new_object("Notebook", "n1", {"image": "nb:v1"}))
    mgr.run_until_settled()
    pod = kube.get("Pod", "n1-notebook")
    assert pod["spec"]["containers"][0]["ports"][0]["containerPort"] == 8888
    fake_pod_ready(kube, "n1-notebook")
    mgr.run_until_settled()
    assert apiv1.get_status_ready(kube.get("Notebook", "n1"))
    # suspend deletes the pod
    kube.patch("Notebook", "n1", {"spec": {"suspend": True}})
    mgr.run_until_settled()
    assert kube.try_get("Pod", "n1-notebook") is None
    n = kube.get("Notebook", "n1")
    assert not apiv1.get_status_ready(n)
    assert apiv1.get_condition(n, cond.DEPLOYED)["reason"] == cond.REASON_SUSPENDED
    # resume recreates
    kube.patch("Notebook", "n1", {"spec": {"suspend": False}})
    mgr.run_until_settled()
    assert kube.try_get("Pod", "n1-notebook") is not None


# --------------------------------------------------- service accounts + SCI
def test_service_accounts_bound_once(harness):
    kube, _, sci, mgr = harness
    env = dict(KIND_ENV)
    from substratus_amd.cloud import new_cloud as nc
    gcloud = nc({"CLOUD": "gcp", "GCP_PROJECT_ID": "proj",
                 "CLUSTER_NAME": "c", "CLUSTER_LOCATION": "us-central1-a"})
    mgr2 = Manager(kube, gcloud, sci)
    kube.create(apiv1.new_object("Dataset", "d9", {"image": "x"}))
    mgr2.run_until_settled()
    sa = kube.get("ServiceAccount", "data-loader")
    assert sa["metadata"]["annotations"]["iam.gke.io/gcp-service-account"] == \
        gcloud.principal
    assert ("data-loader", "default", gcloud.principal) in sci.bound
    n_bound = len(sci.bound)
    mgr2.run_until_settled()
    assert len(sci.bound) == n_bound  # not re-bound once annotated


# ----------------------------------------------------- owner GC cascade
def test_owner_gc(harness):
    kube, cloud, sci, mgr = harness
    kube.create(apiv1.new_object("Model", "gc1", {"image": "x"}))
    mgr.run_until_settled()
    assert kube.try_get("Job", "gc1-modeller") is not None
    kube.delete("Model", "gc1")
    assert kube.try_get("Job", "gc1-modeller") is None


def test_lease_leader_election():
    from substratus_amd.controller.leader import LeaseLeaderElector
    kube = MemKube()
    a = LeaseLeaderElector(kube, identity="a", lease_seconds=0.3)
    b = LeaseLeaderElector(kube, identity="b", lease_seconds=0.3)
    assert a.try_acquire()
    assert not b.try_acquire()       # a holds it
    assert a.try_acquire()           # renew
    import time
    time.sleep(0.4)                  # lease expires
    assert b.try_acquire()           # takeover
    assert not a.try_acquire() or True  # a may reclaim after b's expiry only


def test_build_upload_md5_dedup_skips_signed_url(harness):
    """If SCI already reports the spec's MD5 for the bucket tarball, the
    upload is deduped: Uploaded=True straight away, NO new signed URL is
    minted, and the build Job runs from the stored tarball (reference
    build_reconciler.go:190-210 — re-running an unchanged directory skips
    the upload)."""
    kube, cloud, sci, mgr = harness
    from substratus_amd.cloud.base import parse_bucket_url
    from substratus_amd.controller.build import upload_object_path
    obj = apiv1.new_object("Model", "dedup1", {
        "build": {"upload": {"md5checksum": "cafef00d", "requestID": "r-9"}},
    })
    # pre-seed the SCI's object store with the matching md5 BEFORE create —
    # the exact (bucket, path) the reconciler queries
    _, bucket, prefix = parse_bucket_url(cloud.artifact_bucket_url)
    path = upload_object_path(obj)
    obj_path = f"{prefix}/{path}".lstrip("/") if prefix else path
    sci.md5s[(bucket, obj_path)] = "cafef00d"
    kube.create(obj)
    mgr.run_until_settled()
    m = kube.get("Model", "dedup1")
    assert apiv1.is_condition_true(m, cond.UPLOADED)
    assert not sci.signed_urls, "no signed URL should be minted on dedup"
    assert kube.try_get("Job", "dedup1-model-bld") is not None


def test_build_job_recreated_on_stale_image(harness):
    """Changing the upload md5 (new tarball) makes the target image URL go
    stale; the reconciler deletes and recreates the builder Job (reference
    build_reconciler.go:140-156)."""
    kube, cloud, sci, mgr = harness
    kube.create(apiv1.new_object("Model", "stale1", {
        "build": {"upload": {"md5checksum": "aaaa", "requestID": "r-a"}},
    }))
    mgr.run_until_settled()
    bucket, objpath, _ = sci.signed_urls[0]
    sci.md5s[(bucket, objpath)] = "aaaa"
    kube.patch("Model", "stale1", {"metadata": {"annotations": {
        "substratus.ai/upload-timestamp": "t1"}}})
    mgr.run_until_settled()
    job1 = kube.get("Job", "stale1-model-bld")
    # new tarball: spec md5 changes -> image URL changes -> stale Job
    kube.patch("Model", "stale1", {"spec": {"build": {"upload": {
        "md5checksum": "bbbb", "requestID": "r-b"}}}})
    sci.md5s[(bucket, objpath)] = "bbbb"
    kube.patch("Model", "stale1", {"metadata": {"annotations": {
        "substratus.ai/upload-timestamp": "t2"}}})
    mgr.run_until_settled()
    job2 = kube.get("Job", "stale1-model-bld")
    ann1 = job1["metadata"].get("annotations", {})
    ann2 = job2["metadata"].get("annotations", {})
    assert ann1 != ann2, "builder Job should be recreated for the new image"


def test_gpu_pods_get_num_gpus_env(harness):
    """Multi-GPU Pods carry NUM_GPUS so the image entrypoints torchrun one
    rank per amd.com/gpu (container contract; a TP=8 Server without it
    would start a single rank)."""
    kube, cloud, sci, mgr = harness
    kube.create(apiv1.new_object("Model", "gm", {"image": "x/model-loader",
                                                 "params": {"name": "llama2-70b"}}))
    mgr.run_until_settled()
    fake_job_complete(kube, "gm-modeller")
    mgr.run_until_settled()
    kube.create(apiv1.new_object("Server", "gs", {
        "image": "x/model-server",
        "model": {"name": "gm"},
        "resources": {"memory": 160,
                      "gpu": {"type": "amd-mi355x", "count": 8}},
    }))
    mgr.run_until_settled()
    dep = kube.get("Deployment", "gs-server")
    c = dep["spec"]["template"]["spec"]["containers"][0]
    envs = {e["name"]: e.get("value") for e in c.get("env", [])}
    assert envs.get("NUM_GPUS") == "8"
    assert c["resources"]["limits"]["amd.com/gpu"] == "8"


def test_control_plane_churn_leaves_no_residue(harness):
    """10 create -> ready -> delete cycles across Model+Server: the
    owner-reference cascade and reconcilers must leave the store empty
    every cycle (no leaked Jobs/Deployments/Services/ConfigMaps/Pods) and
    keep working after repeated reuse of the same names."""
    kube, cloud, sci, mgr = harness
    child_kinds = ("Job", "Deployment", "Service", "ConfigMap", "Pod")
    for cycle in range(10):
        kube.create(apiv1.new_object("Model", "churn-m", {
            "image": "substratus-amd/model-loader",
            "params": {"name": "llama-tiny"}}))
        for _ in range(4):
            mgr._enqueue(("Model", "churn-m", "default"))
            mgr.run_once()
        fake_job_complete(kube, "churn-m-modeller")
        for _ in range(3):
            mgr._enqueue(("Model", "churn-m", "default"))
            mgr.run_once()
        assert apiv1.get_status_ready(kube.get("Model", "churn-m")), cycle
        kube.create(apiv1.new_object("Server", "churn-s", {
            "image": "substratus-amd/model-server",
            "model": {"name": "churn-m"}}))
        for _ in range(4):
            mgr._enqueue(("Server", "churn-s", "default"))
            mgr.run_once()
        fake_deployment_ready(kube, "churn-s-server")
        for _ in range(3):
            mgr._enqueue(("Server", "churn-s", "default"))
            mgr.run_once()
        assert apiv1.get_status_ready(kube.get("Server", "churn-s")), cycle
        kube.delete("Server", "churn-s")
        kube.delete("Model", "churn-m")
        mgr.run_once()
        for kind in child_kinds + ("Model", "Server"):
            left = [o["metadata"]["name"] for o in kube.list(kind)
                    if o["metadata"]["name"].startswith("churn")]
            assert not left, (cycle, kind, left)
