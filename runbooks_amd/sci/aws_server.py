"""SCI server for AWS (S3 presigned PUT + EKS IRSA identity binding).

Parity: reference internal/sci/aws/server.go + cmd/sci-aws/main.go (port
10081). The reference uses aws-sdk-go; here the S3 SigV4 presigner is
implemented directly (hmac/hashlib — no AWS SDK in this image), which also
makes it unit-testable against SigV4 golden values offline.

- CreateSignedURL → presigned PUT with Content-MD5 baked into SignedHeaders
  (server.go:60-86)
- GetObjectMd5    → HEAD object, md5 from the ETag (server.go:36-58)
- BindIdentity    → add an sts:AssumeRoleWithWebIdentity statement for the
  KSA to the IAM role trust policy (server.go:88-162)
"""
from __future__ import annotations

import argparse
import datetime
import hashlib
import hmac
import os
import urllib.parse
import urllib.request
from typing import Callable, Optional

from . import proto
from .server import ControllerServicer, serve


def _sign(key: bytes, msg: str) -> bytes:
    return hmac.new(key, msg.encode(), hashlib.sha256).digest()


def _signing_key(secret: str, date: str, region: str, service: str) -> bytes:
    k = _sign(("AWS4" + secret).encode(), date)
    k = _sign(k, region)
    k = _sign(k, service)
    return _sign(k, "aws4_request")


def presign_put(bucket: str, key: str, *, access_key: str, secret_key: str,
                region: str, expires: int, md5_b64: str = "",
                session_token: str = "",
                now: Optional[datetime.datetime] = None) -> str:
    """SigV4 query-string presigned PUT URL for s3.{region}.amazonaws.com.

    Matches the URL shape the reference's aws-sdk presigner emits for
    PutObjectInput{Bucket, Key, ContentMD5} (reference aws/server.go:60-86).
    """
    now = now or datetime.datetime.now(datetime.timezone.utc)
    amz_date = now.strftime("%Y%m%dT%H%M%SZ")
    datestamp = now.strftime("%Y%m%d")
    host = f"{bucket}.s3.{region}.amazonaws.com"
    canonical_uri = "/" + urllib.parse.quote(key)
    scope = f"{datestamp}/{region}/s3/aws4_request"

    headers = {"host": host}
    if md5_b64:
        headers["content-md5"] = md5_b64
    signed_headers = ";".join(sorted(headers))

    q = {
        "X-Amz-Algorithm": "AWS4-HMAC-SHA256",
        "X-Amz-Credential": f"{access_key}/{scope}",
        "X-Amz-Date": amz_date,
        "X-Amz-Expires": str(expires),
        "X-Amz-SignedHeaders": signed_headers,
    }
    if session_token:
        q["X-Amz-Security-Token"] = session_token
    canonical_query = "&".join(
        f"{urllib.parse.quote(k, safe='')}={urllib.parse.quote(v, safe='')}"
        for k, v in sorted(q.items()))
    canonical_headers = "".join(f"{k}:{headers[k]}\n" for k in sorted(headers))
    canonical_request = "\n".join([
        "PUT", canonical_uri, canonical_query, canonical_headers,
        signed_headers, "UNSIGNED-PAYLOAD"])
    string_to_sign = "\n".join([
        "AWS4-HMAC-SHA256", amz_date, scope,
        hashlib.sha256(canonical_request.encode()).hexdigest()])
    sig = hmac.new(_signing_key(secret_key, datestamp, region, "s3"),
                   string_to_sign.encode(), hashlib.sha256).hexdigest()
    return (f"https://{host}{canonical_uri}?{canonical_query}"
            f"&X-Amz-Signature={sig}")


def irsa_trust_statement(oidc_provider_arn: str, oidc_provider: str,
                         namespace: str, ksa: str) -> dict:
    """Trust-policy statement granting the KSA web-identity assume-role
    (reference aws/server.go:101-142)."""
    return {
        "Effect": "Allow",
        "Principal": {"Federated": oidc_provider_arn},
        "Action": "sts:AssumeRoleWithWebIdentity",
        "Condition": {"StringEquals": {
            f"{oidc_provider}:sub":
                f"system:serviceaccount:{namespace}:{ksa}",
            f"{oidc_provider}:aud": "sts.amazonaws.com",
        }},
    }


class AWSSCI(ControllerServicer):
    def __init__(self, *, region: str = "", access_key: str = "",
                 secret_key: str = "", session_token: str = "",
                 oidc_provider_arn: str = "",
                 http_head: Optional[Callable[[str], dict]] = None,
                 iam_update: Optional[Callable[[str, dict], None]] = None):
        env = os.environ
        self.region = region or env.get("AWS_REGION", "us-west-2")
        self.access_key = access_key or env.get("AWS_ACCESS_KEY_ID", "")
        self.secret_key = secret_key or env.get("AWS_SECRET_ACCESS_KEY", "")
        self.session_token = session_token or env.get("AWS_SESSION_TOKEN", "")
        self.oidc_provider_arn = oidc_provider_arn or \
            env.get("OIDC_PROVIDER_ARN", "")
        self._http_head = http_head or self._default_head
        self._iam_update = iam_update

    def _default_head(self, url: str) -> dict:
        req = urllib.request.Request(url, method="HEAD")
        with urllib.request.urlopen(req) as resp:
            return dict(resp.headers)

    def CreateSignedURL(self, request, context):
        md5_b64 = ""
        if request.md5_checksum:
            md5_b64 = __import__("base64").b64encode(
                bytes.fromhex(request.md5_checksum)).decode()
        url = presign_put(request.bucket_name, request.object_name,
                          access_key=self.access_key,
                          secret_key=self.secret_key, region=self.region,
                          expires=int(request.expiration_seconds) or 300,
                          md5_b64=md5_b64, session_token=self.session_token)
        return proto.CreateSignedURLResponse(url=url)

    def GetObjectMd5(self, request, context):
        # Single-part uploads: ETag == hex md5 (reference aws/server.go:36-58).
        url = presign_put(request.bucket_name, request.object_name,
                          access_key=self.access_key,
                          secret_key=self.secret_key, region=self.region,
                          expires=60)
        headers = self._http_head(url.split("?")[0])
        etag = headers.get("ETag", headers.get("Etag", "")).strip('"')
        return proto.GetObjectMd5Response(md5_checksum=etag)

    def BindIdentity(self, request, context):
        if self._iam_update is None:
            raise NotImplementedError(
                "BindIdentity needs IAM credentials (deploy-time only)")
        oidc_provider = self.oidc_provider_arn.split("/", 1)[-1]
        stmt = irsa_trust_statement(self.oidc_provider_arn, oidc_provider,
                                    request.kubernetes_namespace,
                                    request.kubernetes_service_account)
        self._iam_update(request.principal, stmt)
        return proto.BindIdentityResponse()


def main():
    p = argparse.ArgumentParser(description="SCI server for AWS")
    p.add_argument("--port", type=int, default=10081)
    args = p.parse_args()
    server = serve(AWSSCI(), f"0.0.0.0:{args.port}")
    print(f"sci-aws: grpc :{args.port}", flush=True)
    server.wait_for_termination()


if __name__ == "__main__":
    main()
