"""GCP SCI (credential-gated, like the reference's cloud SCI tests —
internal/sci/gcp/manager_test.go:20-27 skips without
GOOGLE_APPLICATION_CREDENTIALS).

Reference internal/sci/gcp/manager.go:50-204: V4 signed PUT URLs via IAM
SignBlob, object MD5 from attrs, BindIdentity adds
roles/iam.workloadIdentityUser to the GSA IAM policy. The google-cloud
libraries are not installed in this offline image; constructing the service
without them raises with a clear message.
"""

from __future__ import annotations

import base64
import os
from typing import Optional

from . import proto


class GCPSCIService:
    def __init__(self, project: Optional[str] = None):
        try:
            from google.cloud import storage  # type: ignore
            from google.auth import default  # type: ignore
        except ImportError as e:
            raise RuntimeError(
                "GCP SCI needs google-cloud-storage + google-auth "
                "(unavailable in this offline image); deploy the sci-gcp "
                "container with them installed") from e
        creds, detected = default()
        self.project = project or detected
        self.storage = storage.Client(project=self.project,
                                      credentials=creds)

    def CreateSignedURL(self, request, context):
        bucket = self.storage.bucket(request.bucketName)
        blob = bucket.blob(request.objectName)
        kwargs = dict(version="v4", method="PUT",
                      expiration=int(request.expirationSeconds))
        if request.md5Checksum:
            kwargs["content_md5"] = base64.b64encode(
                bytes.fromhex(request.md5Checksum)).decode()
        return proto.CreateSignedURLResponse(
            url=blob.generate_signed_url(**kwargs))

    def GetObjectMd5(self, request, context):
        bucket = self.storage.bucket(request.bucketName)
        blob = bucket.get_blob(request.objectName)
        if blob is None or blob.md5_hash is None:
            import grpc
            context.set_code(grpc.StatusCode.NOT_FOUND)
            return proto.GetObjectMd5Response()
        md5_hex = base64.b64decode(blob.md5_hash).hex()
        return proto.GetObjectMd5Response(md5Checksum=md5_hex)

    def BindIdentity(self, request, context):
        # roles/iam.workloadIdentityUser member on the GSA policy
        # (reference manager.go:148-188)
        from googleapiclient import discovery  # type: ignore
        iam = discovery.build("iam", "v1")
        gsa = request.principal
        resource = f"projects/{self.project}/serviceAccounts/{gsa}"
        member = (f"serviceAccount:{self.project}.svc.id.goog"
                  f"[{request.kubernetesNamespace}/"
                  f"{request.kubernetesServiceAccount}]")
        policy = iam.projects().serviceAccounts().getIamPolicy(
            resource=resource).execute()
        bindings = policy.setdefault("bindings", [])
        for b in bindings:
            if b["role"] == "roles/iam.workloadIdentityUser":
                if member not in b["members"]:
                    b["members"].append(member)
                break
        else:
            bindings.append({"role": "roles/iam.workloadIdentityUser",
                             "members": [member]})
        iam.projects().serviceAccounts().setIamPolicy(
            resource=resource, body={"policy": policy}).execute()
        return proto.BindIdentityResponse()
