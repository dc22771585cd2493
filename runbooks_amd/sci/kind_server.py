"""SCI server for local kind clusters.

Parity: reference internal/sci/kind/server.go + cmd/sci-kind/main.go.
- CreateSignedURL returns a fake "signed" URL pointing at this server's own
  HTTP handler (server.go:83-89; NodePort 30080 in the install manifests).
- The HTTP handler accepts the PUT, checks Content-MD5, writes the file and
  a sidecar md5.txt (server.go:27-81).
- GetObjectMd5 reads the sidecar (server.go:91-106).
- BindIdentity is a no-op (server.go:108-110).
"""
from __future__ import annotations

import argparse
import base64
import hashlib
import os
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

from . import proto
from .server import ControllerServicer, serve


class KindSCI(ControllerServicer):
    def __init__(self, signed_url_address: str = "http://localhost:30080",
                 root: str = "/"):
        self.signed_url_address = signed_url_address.rstrip("/")
        self.root = root

    def CreateSignedURL(self, request, context):
        return proto.CreateSignedURLResponse(
            url=f"{self.signed_url_address}/{request.object_name}")

    def GetObjectMd5(self, request, context):
        path = os.path.join(self.root,
                            os.path.dirname(request.object_name), "md5.txt")
        with open(path) as f:
            return proto.GetObjectMd5Response(md5_checksum=f.read().strip())

    def BindIdentity(self, request, context):
        return proto.BindIdentityResponse()


class _UploadHandler(BaseHTTPRequestHandler):
    sci: KindSCI = None  # set by make_http_server

    def do_PUT(self):
        if self.headers.get("Content-Type") != "application/octet-stream":
            self.send_response(400)
            self.end_headers()
            return
        md5_b64 = self.headers.get("Content-MD5", "")
        if not md5_b64:
            self.send_response(400)
            self.end_headers()
            return
        try:
            md5_hex = base64.b64decode(md5_b64).hex()
        except Exception:
            self.send_response(400)
            self.end_headers()
            return
        length = int(self.headers.get("Content-Length", 0))
        body = self.rfile.read(length)
        if hashlib.md5(body).hexdigest() != md5_hex:
            self.send_response(400)
            self.end_headers()
            return
        rel = self.path.lstrip("/")
        dst = os.path.join(self.sci.root, rel)
        os.makedirs(os.path.dirname(dst), exist_ok=True)
        with open(os.path.join(os.path.dirname(dst), "md5.txt"), "w") as f:
            f.write(md5_hex)
        with open(dst, "wb") as f:
            f.write(body)
        self.send_response(200)
        self.end_headers()

    def log_message(self, fmt, *args):  # quiet
        pass


def make_http_server(sci: KindSCI, port: int = 8080) -> ThreadingHTTPServer:
    handler = type("Handler", (_UploadHandler,), {"sci": sci})
    httpd = ThreadingHTTPServer(("0.0.0.0", port), handler)
    threading.Thread(target=httpd.serve_forever, daemon=True).start()
    return httpd


def main():
    p = argparse.ArgumentParser(description="SCI server for kind")
    p.add_argument("--grpc-port", type=int, default=10080)
    p.add_argument("--http-port", type=int, default=8080)
    p.add_argument("--signed-url-address",
                   default=os.environ.get("SIGNED_URL_ADDRESS",
                                          "http://localhost:30080"))
    args = p.parse_args()
    sci = KindSCI(signed_url_address=args.signed_url_address)
    make_http_server(sci, args.http_port)
    server = serve(sci, f"0.0.0.0:{args.grpc_port}")
    print(f"sci-kind: grpc :{args.grpc_port} http :{args.http_port}",
          flush=True)
    server.wait_for_termination()


if __name__ == "__main__":
    main()
