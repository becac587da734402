"""Client library for the `sub` CLI (reference internal/client/, 986 LoC Go).

Two backends behind one interface: HTTP against the platform daemon
(`sub up`), or in-process MemKube (tests). Implements the reference's client
behaviors: server-side apply, 1-s WaitReady polling
(internal/client/client.go:114-135), the build-upload dance — tar.gz the
directory (Dockerfile required, internal/client/upload.go:38-68), create the
object with build.upload{md5checksum, requestID}, watch status.buildUpload
for the signed URL matching the requestID, HTTP PUT with the tarball, patch
the upload-timestamp annotation to retrigger reconcile
(upload.go:126-192, 308-351) — and NotebookForObject conversion
(internal/client/notebook.go:21-86).
"""

from __future__ import annotations

import hashlib
import io
import os
import tarfile
import time
import uuid
from typing import Optional

import httpx

from ..api import types as apiv1


class ClientError(RuntimeError):
    pass


class HTTPClient:
    def __init__(self, base_url: str = "http://127.0.0.1:8070"):
        self.http = httpx.Client(base_url=base_url, timeout=30)

    def create(self, obj: dict) -> dict:
        r = self.http.post("/api/objects", json=obj)
        if r.status_code == 409:
            raise ClientError(f"already exists: {obj['metadata']['name']}")
        r.raise_for_status()
        return r.json()

    def apply(self, obj: dict) -> dict:
        r = self.http.put("/api/objects", json=obj)
        r.raise_for_status()
        return r.json()

    def get(self, kind: str, name: str, ns: str = "default") -> Optional[dict]:
        r = self.http.get(f"/api/objects/{kind}/{ns}/{name}")
        if r.status_code == 404:
            return None
        r.raise_for_status()
        return r.json()

    def list(self, kind: str, ns: str = "default") -> list[dict]:
        r = self.http.get(f"/api/objects/{kind}", params={"namespace": ns})
        r.raise_for_status()
        return r.json()["items"]

    def patch(self, kind: str, name: str, patch: dict,
              ns: str = "default") -> dict:
        r = self.http.patch(f"/api/objects/{kind}/{ns}/{name}", json=patch)
        r.raise_for_status()
        return r.json()

    def delete(self, kind: str, name: str, ns: str = "default") -> None:
        r = self.http.delete(f"/api/objects/{kind}/{ns}/{name}")
        if r.status_code == 404:
            raise ClientError(f"{kind}/{name} not found")
        r.raise_for_status()


class InProcClient:
    """Same interface over a MemKube (tests / embedded)."""

    def __init__(self, kube):
        self.kube = kube

    def create(self, obj):
        return self.kube.create(obj)

    def apply(self, obj):
        return self.kube.apply(obj)

    def get(self, kind, name, ns="default"):
        return self.kube.try_get(kind, name, ns)

    def list(self, kind, ns="default"):
        return self.kube.list(kind, ns)

    def patch(self, kind, name, patch, ns="default"):
        return self.kube.patch(kind, name, patch, ns)

    def delete(self, kind, name, ns="default"):
        self.kube.delete(kind, name, ns)


# ------------------------------------------------------------------- helpers
def wait_ready(client, kind: str, name: str, ns: str = "default",
               timeout: float = 720.0, poll: float = 1.0,
               progress=None) -> dict:
    """1-s poll on .status.ready (reference client.go:114-135; the 720 s
    budget is the reference system test's — test/system.sh:53-54)."""
    deadline = time.time() + timeout
    while time.time() < deadline:
        obj = client.get(kind, name, ns)
        if obj is not None:
            if progress:
                progress(obj)
            if apiv1.get_status_ready(obj):
                return obj
        time.sleep(poll)
    raise ClientError(f"{kind}/{name} not ready after {timeout:.0f}s")


def prepare_tarball(directory: str) -> tuple[bytes, str]:
    """tar.gz of the build context; Dockerfile required
    (reference upload.go:38-68)."""
    if not os.path.exists(os.path.join(directory, "Dockerfile")):
        raise ClientError(f"no Dockerfile in {directory}")
    buf = io.BytesIO()
    with tarfile.open(fileobj=buf, mode="w:gz") as tar:
        for root, dirs, files in os.walk(directory):
            dirs[:] = [d for d in dirs if not d.startswith(".")]
            for f in files:
                full = os.path.join(root, f)
                tar.add(full, arcname=os.path.relpath(full, directory))
    data = buf.getvalue()
    return data, hashlib.md5(data).hexdigest()


def upload_build(client, obj: dict, directory: str,
                 timeout: float = 120.0) -> dict:
    """The reference's upload dance (upload.go:126-192)."""
    data, md5 = prepare_tarball(directory)
    request_id = str(uuid.uuid4())
    obj.setdefault("spec", {})["build"] = {
        "upload": {"md5checksum": md5, "requestID": request_id}}
    obj = client.apply(obj)
    kind, name = obj["kind"], obj["metadata"]["name"]
    ns = obj["metadata"].get("namespace", "default")
    deadline = time.time() + timeout
    while time.time() < deadline:
        cur = client.get(kind, name, ns)
        up = apiv1.get_status_upload(cur or {})
        if up.get("requestID") == request_id and up.get("signedURL"):
            r = httpx.put(up["signedURL"], content=data,
                          headers={"Content-MD5": md5}, timeout=60)
            r.raise_for_status()
            client.patch(kind, name, {"metadata": {"annotations": {
                "substratus.ai/upload-timestamp": str(time.time())}}}, ns)
            return client.get(kind, name, ns)
        time.sleep(0.5)
    raise ClientError("timed out waiting for signed upload URL")


def notebook_for_object(obj: dict) -> dict:
    """Model/Dataset/Server → equivalent Notebook spec (dev-env parity,
    reference internal/client/notebook.go:21-86)."""
    kind = obj["kind"]
    name = obj["metadata"]["name"]
    spec = obj.get("spec", {})
    nb_spec: dict = {}
    for key in ("image", "build", "params", "env", "resources"):
        if spec.get(key) is not None:
            nb_spec[key] = spec[key]
    if kind == "Model":
        if (spec.get("model") or {}).get("name"):
            nb_spec["model"] = spec["model"]
        if (spec.get("dataset") or {}).get("name"):
            nb_spec["dataset"] = spec["dataset"]
    elif kind == "Server":
        if (spec.get("model") or {}).get("name"):
            nb_spec["model"] = spec["model"]
    elif kind == "Dataset":
        pass
    elif kind == "Notebook":
        nb_spec = spec
    return apiv1.new_object("Notebook", f"{name}-nb"[:50], nb_spec,
                            obj["metadata"].get("namespace", "default"))


def increment_name(client, kind: str, name: str, ns: str = "default") -> str:
    """`sub run -i`: name, name-2, name-3 ... (reference
    internal/tui/common.go:158-206)."""
    base = name
    n = 1
    while client.get(kind, name, ns) is not None:
        n += 1
        name = f"{base}-{n}"
    return name
