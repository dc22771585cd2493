"""SCI server tests.

Mirrors the reference: kind SCI tested hermetically with a loopback gRPC
channel + real HTTP PUT (reference internal/sci/kind/server_test.go:23-50);
AWS/GCP signing logic tested as pure functions (the reference skips its
AWS/GCP tests without real credentials, reference
internal/sci/aws/server_test.go:44-60).
"""
import base64
import datetime
import hashlib
import urllib.request

import pytest

from runbooks_amd.sci import ControllerClient, proto
from runbooks_amd.sci.aws_server import irsa_trust_statement, presign_put
from runbooks_amd.sci.gcp_server import canonical_v4_request
from runbooks_amd.sci.kind_server import KindSCI, make_http_server
from runbooks_amd.sci.server import serve


def test_proto_wire_roundtrip():
    r = proto.CreateSignedURLRequest(bucket_name="b", object_name="o/p",
                                     expiration_seconds=300,
                                     md5_checksum="ff" * 16)
    data = r.SerializeToString()
    r2 = proto.CreateSignedURLRequest.FromString(data)
    assert r2.bucket_name == "b" and r2.object_name == "o/p"
    assert r2.expiration_seconds == 300


def test_kind_sci_end_to_end(tmp_path):
    # gRPC loopback + HTTP upload, like the reference's httptest harness.
    sci = KindSCI(signed_url_address="http://127.0.0.1:0", root=str(tmp_path))
    httpd = make_http_server(sci, port=0)
    port = httpd.server_address[1]
    sci.signed_url_address = f"http://127.0.0.1:{port}"

    server = serve(sci, "127.0.0.1:0")
    client = ControllerClient(f"127.0.0.1:{server.bound_port}")

    body = b"tarball-bytes"
    md5_hex = hashlib.md5(body).hexdigest()

    resp = client.create_signed_url("bucket", "abc123/uploads/latest.tar.gz",
                                    300, md5_hex)
    assert resp.url.startswith(f"http://127.0.0.1:{port}/")

    req = urllib.request.Request(resp.url, data=body, method="PUT", headers={
        "Content-Type": "application/octet-stream",
        "Content-MD5": base64.b64encode(bytes.fromhex(md5_hex)).decode(),
    })
    with urllib.request.urlopen(req) as r:
        assert r.status == 200

    got = client.get_object_md5("bucket", "abc123/uploads/latest.tar.gz")
    assert got.md5_checksum == md5_hex

    # BindIdentity is a no-op on kind (reference kind/server.go:108-110)
    client.bind_identity("modeller", "default")
    server.stop(0)
    httpd.shutdown()


def test_kind_http_rejects_bad_md5(tmp_path):
    sci = KindSCI(root=str(tmp_path))
    httpd = make_http_server(sci, port=0)
    port = httpd.server_address[1]
    req = urllib.request.Request(
        f"http://127.0.0.1:{port}/bucket/x/uploads/latest.tar.gz",
        data=b"data", method="PUT", headers={
            "Content-Type": "application/octet-stream",
            "Content-MD5": base64.b64encode(b"0" * 16).decode(),
        })
    with pytest.raises(urllib.error.HTTPError) as e:
        urllib.request.urlopen(req)
    assert e.value.code == 400
    httpd.shutdown()


def test_aws_presign_shape():
    now = datetime.datetime(2026, 1, 2, 3, 4, 5,
                            tzinfo=datetime.timezone.utc)
    url = presign_put("bkt", "p/latest.tar.gz", access_key="AKIDEXAMPLE",
                      secret_key="secret", region="us-west-2", expires=300,
                      md5_b64="AAAA", now=now)
    assert url.startswith("https://bkt.s3.us-west-2.amazonaws.com/"
                          "p/latest.tar.gz?")
    assert "X-Amz-Algorithm=AWS4-HMAC-SHA256" in url
    assert "X-Amz-Date=20260102T030405Z" in url
    assert "X-Amz-SignedHeaders=content-md5%3Bhost" in url
    assert "X-Amz-Signature=" in url
    # deterministic
    assert url == presign_put("bkt", "p/latest.tar.gz",
                              access_key="AKIDEXAMPLE", secret_key="secret",
                              region="us-west-2", expires=300,
                              md5_b64="AAAA", now=now)


def test_aws_trust_statement():
    stmt = irsa_trust_statement(
        "arn:aws:iam::123:oidc-provider/oidc.eks.us-west-2.amazonaws.com/id/X",
        "oidc.eks.us-west-2.amazonaws.com/id/X", "default", "modeller")
    cond = stmt["Condition"]["StringEquals"]
    assert cond["oidc.eks.us-west-2.amazonaws.com/id/X:sub"] == \
        "system:serviceaccount:default:modeller"
    assert stmt["Action"] == "sts:AssumeRoleWithWebIdentity"


def test_gcs_v4_string_to_sign():
    now = datetime.datetime(2026, 1, 2, 3, 4, 5,
                            tzinfo=datetime.timezone.utc)
    sts, query, host_uri = canonical_v4_request(
        "bkt", "h/uploads/latest.tar.gz", "sa@p.iam.gserviceaccount.com",
        expires=300, md5_b64="AAAA", now=now)
    assert sts.startswith("GOOG4-RSA-SHA256\n20260102T030405Z\n"
                          "20260102/auto/storage/goog4_request\n")
    assert "X-Goog-Credential=sa%40p.iam.gserviceaccount.com" in query
    assert host_uri == "storage.googleapis.com/bkt/h/uploads/latest.tar.gz"
