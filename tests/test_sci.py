"""SCI tests mirroring the reference's kind SCI integration test
(internal/sci/kind/server_test.go:24-97): in-process gRPC + HTTP upload→MD5
roundtrip on a tempdir; cloud SCIs are credential-gated like the reference's
(gcp/manager_test.go:20-27)."""

import hashlib
import os

import httpx
import pytest

from substratus_amd.sci.client import GrpcSCIClient
from substratus_amd.sci.kind_server import serve


@pytest.fixture()
def kind_sci(tmp_path):
    import socket
    def free():
        s = socket.socket(); s.bind(("127.0.0.1", 0))
        p = s.getsockname()[1]; s.close(); return p
    gport, hport = free(), free()
    server, httpd = serve(str(tmp_path), grpc_port=gport, http_port=hport,
                          url_host=f"http://127.0.0.1:{hport}", block=False)
    yield GrpcSCIClient(f"localhost:{gport}"), hport, tmp_path
    server.stop(grace=None)
    httpd.shutdown()


def test_signed_url_upload_md5_roundtrip(kind_sci):
    client, hport, bucket = kind_sci
    # object not uploaded yet
    assert client.get_object_md5("b1", "uploads/x.tar.gz") is None
    data = b"hello substratus" * 100
    md5 = hashlib.md5(data).hexdigest()
    url = client.create_signed_url("b1", "uploads/x.tar.gz", 300,
                                   md5_checksum=md5)
    assert url.startswith("http://127.0.0.1")
    r = httpx.put(url, content=data)
    assert r.status_code == 200
    # SCI now reports the stored MD5 (the controller's dedup check)
    assert client.get_object_md5("b1", "uploads/x.tar.gz") == md5
    # the file landed under the bucket dir
    assert (bucket / "b1" / "uploads" / "x.tar.gz").exists()


def test_bind_identity_noop(kind_sci):
    client, _, _ = kind_sci
    client.bind_identity("sa", "default", "principal")  # no-op, no error


def test_grpc_client_against_controllers(kind_sci, tmp_path):
    """The build reconciler's upload flow over the REAL gRPC SCI (not the
    fake): signed URL published, PUT, MD5 dedup, built."""
    client, hport, bucket = kind_sci
    from substratus_amd.api import conditions as cond
    from substratus_amd.api import types as apiv1
    from substratus_amd.cloud import new_cloud
    from substratus_amd.controller.manager import Manager
    from substratus_amd.kube.memapi import MemKube
    kube = MemKube()
    cloud = new_cloud({"CLOUD": "kind", "CLUSTER_NAME": "t",
                       "BUCKET_DIR": str(bucket)})
    mgr = Manager(kube, cloud, client)
    data = b"FROM scratch\n"
    md5 = hashlib.md5(data).hexdigest()
    kube.create(apiv1.new_object("Model", "up", {
        "build": {"upload": {"md5checksum": md5, "requestID": "rq-9"}}}))
    mgr.run_until_settled()
    m = kube.get("Model", "up")
    url = apiv1.get_status_upload(m)["signedURL"]
    assert httpx.put(url, content=data).status_code == 200
    kube.patch("Model", "up", {"metadata": {"annotations": {
        "substratus.ai/upload-timestamp": "t1"}}})
    mgr.run_until_settled()
    m = kube.get("Model", "up")
    assert apiv1.is_condition_true(m, cond.UPLOADED)


def test_cloud_scis_credential_gated():
    if not os.environ.get("GOOGLE_APPLICATION_CREDENTIALS"):
        with pytest.raises(RuntimeError):
            from substratus_amd.sci.gcp_server import GCPSCIService
            GCPSCIService()
    if not os.environ.get("AWS_ACCOUNT_ID"):
        with pytest.raises(RuntimeError):
            from substratus_amd.sci.aws_server import AWSSCIService
            AWSSCIService("123")
