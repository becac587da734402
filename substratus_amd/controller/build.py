"""Generic BuildReconciler, instantiated for Model/Server/Notebook/Dataset
(reference internal/controller/build_reconciler.go:59-580).

State machine:
  * spec.image set directly → Built=True, done.
  * spec.build.git → git-clone initContainer + kaniko build Job; on Job
    completion stamp spec.image with the deterministic registry URL
    (cloud.object_built_image_url); the Job is recreated when the image
    annotation goes stale.
  * spec.build.upload{md5checksum, requestID} → signed-URL dance:
      1. ask SCI for the stored object's MD5; if it equals the spec's, the
         tarball is already in the bucket → Uploaded=True, run the build Job
         from the bucket tarball.
      2. otherwise mint a signed PUT URL via SCI and publish it in
         status.buildUpload {signedURL, requestID, expiration}; the CLI
         watches for its requestID, PUTs the tarball (Content-MD5), then
         patches the upload-timestamp annotation to retrigger reconcile
         (reference internal/client/upload.go:126-192).
"""

from __future__ import annotations

import time

from ..api import conditions as cond
from ..api import types as apiv1
from ..cloud.base import Cloud, parse_bucket_url
from ..cloud.resources import container_builder_resources
from ..kube.memapi import MemKube
from .service_accounts import CONTAINER_BUILDER_SA
from .utils import Result, job_result, owned, reconcile_job

SIGNED_URL_EXPIRY_SECONDS = 300
UPLOAD_TS_ANNOTATION = "substratus.ai/upload-timestamp"
IMAGE_ANNOTATION = "substratus.ai/image"


def build_job_name(obj: dict) -> str:
    # {name}-{kind}-bld (reference build_reconciler.go:576-580; suffix <=13)
    return f"{obj['metadata']['name']}-{obj['kind'].lower()}-bld"


def upload_object_path(obj: dict) -> str:
    """Deterministic bucket path of the uploaded build context tarball
    (reference build_reconciler.go:535-574 keys it by namespace/kind/name)."""
    md = obj["metadata"]
    return (f"uploads/{md['namespace']}/{obj['kind'].lower()}s/"
            f"{md['name']}/build-context.tar.gz")


class BuildReconciler:
    def __init__(self, kube: MemKube, cloud: Cloud, sci):
        self.kube = kube
        self.cloud = cloud
        self.sci = sci

    # ------------------------------------------------------------------ main
    def reconcile(self, obj: dict) -> Result:
        """Returns success when the object has a built image; mutates obj's
        spec/status in place (caller persists)."""
        if apiv1.get_image(obj) and not apiv1.get_build(obj):
            apiv1.set_condition(obj, cond.BUILT, True, "ImageSpecified")
            return Result()
        build = apiv1.get_build(obj)
        if not build:
            apiv1.set_condition(obj, cond.BUILT, False, "NoImageOrBuild",
                                "spec.image or spec.build required")
            return Result.wait(30)
        if build.get("upload"):
            res = self._reconcile_upload(obj, build["upload"])
            if not res.success:
                return res
        return self._reconcile_build_job(obj, build)

    # -------------------------------------------------------------- upload
    def _reconcile_upload(self, obj: dict, upload: dict) -> Result:
        md5 = upload.get("md5checksum", "")
        request_id = upload.get("requestID", "")
        if not md5 or not request_id:
            apiv1.set_condition(obj, cond.UPLOADED, False, "InvalidUpload",
                                "build.upload needs md5checksum and requestID")
            return Result.wait(30)
        _, bucket, prefix = parse_bucket_url(self.cloud.artifact_bucket_url)
        path = upload_object_path(obj)
        obj_path = f"{prefix}/{path}".lstrip("/") if prefix else path
        stored = self.sci.get_object_md5(bucket, obj_path)
        if stored == md5:
            # dedup by stored MD5 (reference build_reconciler.go:190-210)
            apiv1.set_condition(obj, cond.UPLOADED, True,
                                cond.REASON_UPLOAD_FOUND)
            return Result()
        status_upload = apiv1.get_status_upload(obj)
        if (status_upload.get("requestID") != request_id or
                _expired(status_upload.get("expiration"))):
            url = self.sci.create_signed_url(bucket, obj_path,
                                             SIGNED_URL_EXPIRY_SECONDS,
                                             md5_checksum=md5)
            apiv1.set_status_upload(obj, {
                "signedURL": url,
                "requestID": request_id,
                "expiration": _iso(time.time() + SIGNED_URL_EXPIRY_SECONDS),
            })
        apiv1.set_condition(obj, cond.UPLOADED, False,
                            cond.REASON_AWAITING_UPLOAD)
        return Result.wait(3)

    # ----------------------------------------------------------- build job
    def _reconcile_build_job(self, obj: dict, build: dict) -> Result:
        image_url = self.cloud.object_built_image_url(obj)
        ns = obj["metadata"].get("namespace", "default")
        name = build_job_name(obj)
        existing = self.kube.try_get("Job", name, ns)
        if existing is not None:
            # recreate when the target image went stale
            # (reference build_reconciler.go:140-156)
            ann = existing["metadata"].get("annotations", {})
            if ann.get(IMAGE_ANNOTATION) != image_url:
                self.kube.delete("Job", name, ns)
                existing = None
        if existing is None:
            job = (self._git_build_job(obj, build, image_url)
                   if build.get("git")
                   else self._storage_build_job(obj, image_url))
            existing = reconcile_job(self.kube, job)
        complete, failed = job_result(existing)
        if failed:
            apiv1.set_condition(obj, cond.BUILT, False,
                                cond.REASON_BUILD_JOB_FAILED)
            return Result.wait(30)
        if not complete:
            reason = (cond.REASON_GIT_BUILD_IN_PROGRESS if build.get("git")
                      else cond.REASON_UPLOAD_BUILD_IN_PROGRESS)
            apiv1.set_condition(obj, cond.BUILT, False, reason)
            return Result.wait(3)
        apiv1.set_image(obj, image_url)
        apiv1.set_condition(obj, cond.BUILT, True, "JobComplete")
        return Result()

    def _job_scaffold(self, obj: dict, image_url: str, containers: list,
                      init_containers: list | None = None) -> dict:
        job = {
            "apiVersion": "batch/v1",
            "kind": "Job",
            "metadata": {
                "name": build_job_name(obj),
                "namespace": obj["metadata"].get("namespace", "default"),
                "annotations": {IMAGE_ANNOTATION: image_url},
            },
            "spec": {
                "backoffLimit": 2,
                "template": {
                    "metadata": {"annotations": {
                        "kubectl.kubernetes.io/default-container": "builder"}},
                    "spec": {
                        "serviceAccountName": CONTAINER_BUILDER_SA,
                        "restartPolicy": "Never",
                        # build Pods run as root
                        # (reference build_reconciler.go:378-382)
                        "securityContext": {"runAsUser": 0},
                        "containers": containers,
                    },
                },
            },
        }
        if init_containers:
            job["spec"]["template"]["spec"]["initContainers"] = init_containers
        res = container_builder_resources(self.cloud.name())
        if res:
            for c in job["spec"]["template"]["spec"]["containers"]:
                c["resources"] = res
        return owned(job, obj)

    def _git_build_job(self, obj: dict, build: dict, image_url: str) -> dict:
        """git-clone initContainer + kaniko builder
        (reference build_reconciler.go:270-403)."""
        git = build["git"]
        clone_args = ["clone", git["url"], "/workspace"]
        if git.get("branch"):
            clone_args += ["--branch", git["branch"]]
        if git.get("tag"):
            clone_args += ["--branch", git["tag"]]
        context_dir = "/workspace"
        if git.get("path"):
            context_dir = f"/workspace/{git['path']}"
        init = [{
            "name": "git-clone",
            "image": "alpine/git",
            "args": clone_args,
            "volumeMounts": [{"name": "workspace", "mountPath": "/workspace"}],
        }]
        builder = [{
            "name": "builder",
            "image": "gcr.io/kaniko-project/executor:latest",
            "args": [f"--context=dir://{context_dir}",
                     f"--destination={image_url}", "--cache=true"],
            "volumeMounts": [{"name": "workspace", "mountPath": "/workspace"}],
        }]
        job = self._job_scaffold(obj, image_url, builder, init)
        job["spec"]["template"]["spec"]["volumes"] = [
            {"name": "workspace", "emptyDir": {}}]
        return job

    def _storage_build_job(self, obj: dict, image_url: str) -> dict:
        """kaniko from the uploaded bucket tarball
        (reference build_reconciler.go:405-533)."""
        _, bucket, prefix = parse_bucket_url(self.cloud.artifact_bucket_url)
        path = upload_object_path(obj)
        obj_path = f"{prefix}/{path}".lstrip("/") if prefix else path
        scheme = self.cloud.artifact_bucket_url.split("://", 1)[0]
        context = (f"tar:///bucket/{obj_path}" if scheme == "tar"
                   else f"{scheme}://{bucket}/{obj_path}")
        builder = [{
            "name": "builder",
            "image": "gcr.io/kaniko-project/executor:latest",
            "args": [f"--context={context}",
                     f"--destination={image_url}", "--cache=true"],
        }]
        job = self._job_scaffold(obj, image_url, builder)
        if scheme == "tar":  # kind: hostPath bucket
            job["spec"]["template"]["spec"].setdefault("volumes", []).append(
                {"name": "bucket", "hostPath": {"path": "/bucket"}})
            for c in job["spec"]["template"]["spec"]["containers"]:
                c.setdefault("volumeMounts", []).append(
                    {"name": "bucket", "mountPath": "/bucket"})
        return job


def _iso(ts: float) -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime(ts))


def _expired(iso: str | None) -> bool:
    import calendar
    if not iso:
        return True
    try:
        t = calendar.timegm(time.strptime(iso, "%Y-%m-%dT%H:%M:%SZ"))
    except ValueError:
        return True
    return t < time.time() - 1
