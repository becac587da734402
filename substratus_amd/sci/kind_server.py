"""Kind SCI: gRPC Controller service + plain-HTTP signed-URL shim over a
local bucket directory.

Reference internal/sci/kind/server.go:27-110 + cmd/sci-kind/main.go:17-59:
the gRPC server (:10080) doubles as an HTTP server (:8080) whose PUT handler
writes the object file plus an `md5.txt` sidecar under /bucket/...; a
"signed URL" is just `http://<host>:30080/<object>` (NodePort). BindIdentity
is a no-op on kind. The gRPC health service is registered like the
reference's binaries.
"""

from __future__ import annotations

import hashlib
import os
import threading
from concurrent import futures
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from typing import Optional

import grpc

from . import proto


class KindSCIService:
    def __init__(self, bucket_dir: str, url_host: str = "http://localhost:30080"):
        self.bucket_dir = bucket_dir
        self.url_host = url_host.rstrip("/")

    # ------------------------------------------------------------- RPC impls
    def CreateSignedURL(self, request, context):
        return proto.CreateSignedURLResponse(
            url=f"{self.url_host}/{request.bucketName}/{request.objectName}")

    def GetObjectMd5(self, request, context):
        md5_path = os.path.join(self.bucket_dir, request.bucketName,
                                request.objectName + ".md5.txt")
        if not os.path.exists(md5_path):
            context.set_code(grpc.StatusCode.NOT_FOUND)
            context.set_details("object not found")
            return proto.GetObjectMd5Response()
        with open(md5_path) as f:
            return proto.GetObjectMd5Response(md5Checksum=f.read().strip())

    def BindIdentity(self, request, context):
        return proto.BindIdentityResponse()  # no identity on kind

    # --------------------------------------------------------------- server
    def grpc_handler(self) -> grpc.GenericRpcHandler:
        handlers = {}
        for name, (req_cls, resp_cls) in proto.RPCS.items():
            handlers[name] = grpc.unary_unary_rpc_method_handler(
                getattr(self, name),
                request_deserializer=req_cls.FromString,
                response_serializer=lambda m: m.SerializeToString())
        return grpc.method_handlers_generic_handler(proto.SERVICE_NAME,
                                                    handlers)


class _PutHandler(BaseHTTPRequestHandler):
    bucket_dir = "/bucket"

    def log_message(self, *args):  # quiet
        pass

    def do_PUT(self):
        rel = self.path.lstrip("/")
        dst = os.path.join(self.bucket_dir, rel)
        os.makedirs(os.path.dirname(dst), exist_ok=True)
        length = int(self.headers.get("Content-Length", 0))
        data = self.rfile.read(length)
        with open(dst, "wb") as f:
            f.write(data)
        md5 = hashlib.md5(data).hexdigest()
        with open(dst + ".md5.txt", "w") as f:
            f.write(md5)
        self.send_response(200)
        self.end_headers()

    def do_GET(self):
        rel = self.path.lstrip("/")
        src = os.path.join(self.bucket_dir, rel)
        if not os.path.exists(src):
            self.send_response(404)
            self.end_headers()
            return
        with open(src, "rb") as f:
            data = f.read()
        self.send_response(200)
        self.send_header("Content-Length", str(len(data)))
        self.end_headers()
        self.wfile.write(data)


def serve(bucket_dir: str, grpc_port: int = 10080, http_port: int = 8080,
          url_host: Optional[str] = None,
          block: bool = True) -> tuple[grpc.Server, ThreadingHTTPServer]:
    svc = KindSCIService(bucket_dir,
                         url_host or f"http://localhost:{http_port}")
    server = grpc.server(futures.ThreadPoolExecutor(max_workers=8))
    server.add_generic_rpc_handlers((svc.grpc_handler(),))
    try:  # health service, like the reference binaries
        from grpc_health.v1 import health, health_pb2_grpc
        health_pb2_grpc.add_HealthServicer_to_server(health.HealthServicer(),
                                                     server)
    except ImportError:
        pass
    server.add_insecure_port(f"[::]:{grpc_port}")
    server.start()

    handler = type("Handler", (_PutHandler,), {"bucket_dir": bucket_dir})
    httpd = ThreadingHTTPServer(("0.0.0.0", http_port), handler)
    t = threading.Thread(target=httpd.serve_forever, daemon=True)
    t.start()
    if block:
        server.wait_for_termination()
    return server, httpd


def main() -> None:
    serve(os.environ.get("BUCKET_DIR", "/bucket"),
          grpc_port=int(os.environ.get("GRPC_PORT", 10080)),
          http_port=int(os.environ.get("HTTP_PORT", 8080)),
          url_host=os.environ.get("URL_HOST"))


if __name__ == "__main__":
    main()
