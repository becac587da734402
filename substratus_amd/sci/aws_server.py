"""AWS SCI (credential-gated — reference internal/sci/aws/server_test.go:
65-80 skips without AWS_ACCOUNT_ID).

Reference internal/sci/aws/server.go:36-201: S3 presigned PUT with
Content-MD5, MD5 from the ETag, BindIdentity patches the IAM role trust
policy with an OIDC sts:AssumeRoleWithWebIdentity condition. boto3 is not in
this offline image; constructing without it raises clearly.
"""

from __future__ import annotations

import base64
import json
from typing import Optional

from . import proto


class AWSSCIService:
    def __init__(self, account_id: str, oidc_provider_url: str = ""):
        try:
            import boto3  # type: ignore
        except ImportError as e:
            raise RuntimeError(
                "AWS SCI needs boto3 (unavailable in this offline image); "
                "deploy the sci-aws container with it installed") from e
        self.account_id = account_id
        self.oidc = oidc_provider_url.removeprefix("https://")
        self.s3 = boto3.client("s3")
        self.iam = boto3.client("iam")

    def CreateSignedURL(self, request, context):
        params = {"Bucket": request.bucketName, "Key": request.objectName}
        if request.md5Checksum:
            params["ContentMD5"] = base64.b64encode(
                bytes.fromhex(request.md5Checksum)).decode()
        url = self.s3.generate_presigned_url(
            "put_object", Params=params,
            ExpiresIn=int(request.expirationSeconds))
        return proto.CreateSignedURLResponse(url=url)

    def GetObjectMd5(self, request, context):
        try:
            head = self.s3.head_object(Bucket=request.bucketName,
                                       Key=request.objectName)
        except self.s3.exceptions.ClientError:
            import grpc
            context.set_code(grpc.StatusCode.NOT_FOUND)
            return proto.GetObjectMd5Response()
        etag = head["ETag"].strip('"')
        return proto.GetObjectMd5Response(md5Checksum=etag)

    def BindIdentity(self, request, context):
        role_name = request.principal.rsplit("/", 1)[-1]
        sub = (f"system:serviceaccount:{request.kubernetesNamespace}:"
               f"{request.kubernetesServiceAccount}")
        role = self.iam.get_role(RoleName=role_name)
        policy = role["Role"]["AssumeRolePolicyDocument"]
        statement = {
            "Effect": "Allow",
            "Principal": {"Federated":
                          f"arn:aws:iam::{self.account_id}:oidc-provider/"
                          f"{self.oidc}"},
            "Action": "sts:AssumeRoleWithWebIdentity",
            "Condition": {"StringEquals": {f"{self.oidc}:sub": sub}},
        }
        stmts = policy.setdefault("Statement", [])
        if statement not in stmts:
            stmts.append(statement)
        self.iam.update_assume_role_policy(
            RoleName=role_name, PolicyDocument=json.dumps(policy))
        return proto.BindIdentityResponse()
