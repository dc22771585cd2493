"""SCI server for GCP (GCS V4 signed URLs + workload-identity binding).

Parity: reference internal/sci/gcp/manager.go + cmd/sci-gcp/main.go.
The reference signs with IAMCredentials SignBlob so no private key is
mounted (manager.go:50-104); this implementation does the same over the
iamcredentials REST API with a metadata-server access token. GetObjectMd5
uses the GCS JSON API's md5Hash (manager.go:106-116); BindIdentity does the
get-modify-set IAM dance adding roles/iam.workloadIdentityUser
(manager.go:118-144).
"""
from __future__ import annotations

import argparse
import base64
import datetime
import hashlib
import json
import os
import urllib.parse
import urllib.request
from typing import Callable, Optional

from . import proto
from .server import ControllerServicer, serve

_METADATA_TOKEN_URL = ("http://metadata.google.internal/computeMetadata/v1/"
                       "instance/service-accounts/default/token")


def _default_token() -> str:
    req = urllib.request.Request(_METADATA_TOKEN_URL,
                                 headers={"Metadata-Flavor": "Google"})
    with urllib.request.urlopen(req, timeout=5) as resp:
        return json.loads(resp.read())["access_token"]


def canonical_v4_request(bucket: str, obj: str, sa_email: str, *,
                         expires: int, md5_b64: str,
                         now: datetime.datetime) -> tuple[str, str, str]:
    """Build the GCS V4 string-to-sign for a PUT. Returns
    (string_to_sign, canonical_query, host). Pure function → unit-testable
    without credentials."""
    amz_date = now.strftime("%Y%m%dT%H%M%SZ")
    datestamp = now.strftime("%Y%m%d")
    host = "storage.googleapis.com"
    uri = f"/{bucket}/" + urllib.parse.quote(obj)
    scope = f"{datestamp}/auto/storage/goog4_request"
    headers = {"host": host}
    if md5_b64:
        headers["content-md5"] = md5_b64
    signed_headers = ";".join(sorted(headers))
    q = {
        "X-Goog-Algorithm": "GOOG4-RSA-SHA256",
        "X-Goog-Credential": f"{sa_email}/{scope}",
        "X-Goog-Date": amz_date,
        "X-Goog-Expires": str(expires),
        "X-Goog-SignedHeaders": signed_headers,
    }
    canonical_query = "&".join(
        f"{urllib.parse.quote(k, safe='')}={urllib.parse.quote(v, safe='')}"
        for k, v in sorted(q.items()))
    canonical_headers = "".join(f"{k}:{headers[k]}\n" for k in sorted(headers))
    canonical_request = "\n".join([
        "PUT", uri, canonical_query, canonical_headers, signed_headers,
        "UNSIGNED-PAYLOAD"])
    string_to_sign = "\n".join([
        "GOOG4-RSA-SHA256", amz_date, scope,
        hashlib.sha256(canonical_request.encode()).hexdigest()])
    return string_to_sign, canonical_query, host + uri


class GCPSCI(ControllerServicer):
    def __init__(self, *, sa_email: str = "", project_id: str = "",
                 token_fn: Optional[Callable[[], str]] = None,
                 http_json: Optional[Callable] = None):
        env = os.environ
        self.sa_email = sa_email or env.get("GSA_EMAIL", "")
        self.project_id = project_id or env.get("PROJECT_ID", "")
        self._token = token_fn or _default_token
        self._http_json = http_json or self._default_http_json

    def _default_http_json(self, method: str, url: str,
                           body: Optional[dict] = None) -> dict:
        data = json.dumps(body).encode() if body is not None else None
        req = urllib.request.Request(url, data=data, method=method, headers={
            "Authorization": f"Bearer {self._token()}",
            "Content-Type": "application/json",
        })
        with urllib.request.urlopen(req, timeout=15) as resp:
            return json.loads(resp.read() or b"{}")

    def _sign_blob(self, payload: bytes) -> bytes:
        """IAMCredentials signBlob (reference manager.go:63-84)."""
        url = (f"https://iamcredentials.googleapis.com/v1/projects/-/"
               f"serviceAccounts/{self.sa_email}:signBlob")
        out = self._http_json("POST", url, {
            "payload": base64.b64encode(payload).decode()})
        return base64.b64decode(out["signedBlob"])

    def CreateSignedURL(self, request, context):
        md5_b64 = ""
        if request.md5_checksum:
            md5_b64 = base64.b64encode(
                bytes.fromhex(request.md5_checksum)).decode()
        sts, query, host_uri = canonical_v4_request(
            request.bucket_name, request.object_name, self.sa_email,
            expires=int(request.expiration_seconds) or 300, md5_b64=md5_b64,
            now=datetime.datetime.now(datetime.timezone.utc))
        sig = self._sign_blob(sts.encode()).hex()
        return proto.CreateSignedURLResponse(
            url=f"https://{host_uri}?{query}&X-Goog-Signature={sig}")

    def GetObjectMd5(self, request, context):
        obj = urllib.parse.quote(request.object_name, safe="")
        url = (f"https://storage.googleapis.com/storage/v1/b/"
               f"{request.bucket_name}/o/{obj}?fields=md5Hash")
        out = self._http_json("GET", url)
        md5_hex = base64.b64decode(out.get("md5Hash", "")).hex()
        return proto.GetObjectMd5Response(md5_checksum=md5_hex)

    def BindIdentity(self, request, context):
        """get-modify-set IAM policy on the GSA adding
        roles/iam.workloadIdentityUser for the KSA member
        (reference manager.go:118-144)."""
        member = (f"serviceAccount:{self.project_id}.svc.id.goog"
                  f"[{request.kubernetes_namespace}/"
                  f"{request.kubernetes_service_account}]")
        base = (f"https://iam.googleapis.com/v1/projects/{self.project_id}/"
                f"serviceAccounts/{request.principal or self.sa_email}")
        policy = self._http_json("POST", f"{base}:getIamPolicy")
        role = "roles/iam.workloadIdentityUser"
        for b in policy.setdefault("bindings", []):
            if b.get("role") == role:
                if member not in b.setdefault("members", []):
                    b["members"].append(member)
                break
        else:
            policy["bindings"].append({"role": role, "members": [member]})
        self._http_json("POST", f"{base}:setIamPolicy", {"policy": policy})
        return proto.BindIdentityResponse()


def main():
    p = argparse.ArgumentParser(description="SCI server for GCP")
    p.add_argument("--port", type=int, default=10080)
    args = p.parse_args()
    server = serve(GCPSCI(), f"0.0.0.0:{args.port}")
    print(f"sci-gcp: grpc :{args.port}", flush=True)
    server.wait_for_termination()


if __name__ == "__main__":
    main()
