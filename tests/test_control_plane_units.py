"""Unit tests for the control plane.

Mirrors the reference's pure-Go unit suites:
- cloud naming/hashing golden values (reference internal/cloud/common_test.go,
  common_internal_test.go)
- resolveEnv secret-ref table tests (reference internal/controller/utils_test.go)
- resources application (reference internal/resources/resources_test.go)
"""
import hashlib
import os

import pytest

from runbooks_amd import resources as res
from runbooks_amd.api import Build, BuildGit, BuildUpload, Model, Resources
from runbooks_amd.api.types import GPUResources, object_from_manifest
from runbooks_amd.cloud import GCP, Kind, new_cloud, parse_bucket_url
from runbooks_amd.cloud.base import object_hash_input
from runbooks_amd.controller.utils import resolve_env


def kind_cloud():
    return new_cloud({"CLOUD": "kind", "CLUSTER_NAME": "kind",
                      "REGISTRY_PORT_5000_TCP_ADDR": "10.4.0.8"})


def gcp_cloud():
    return new_cloud({
        "CLOUD": "gcp", "CLUSTER_NAME": "c1", "PROJECT_ID": "proj",
        "CLUSTER_LOCATION": "us-central1-a", "PRINCIPAL":
        "substratus@proj.iam.gserviceaccount.com"})


def test_image_url_scheme():
    # {registry}/{cluster}-{kind}-{ns}-{name}:{tag}
    # (reference internal/cloud/common.go:18-43)
    c = kind_cloud()
    m = Model(name="m1", namespace="ns1")
    assert c.object_built_image_url(m) == \
        "10.4.0.8:5000/kind-model-ns1-m1:latest"
    m.build = Build(git=BuildGit(url="https://x", tag="v2"))
    assert c.object_built_image_url(m).endswith(":v2")
    m.build = Build(git=BuildGit(url="https://x", branch="dev"))
    assert c.object_built_image_url(m).endswith(":dev")
    m.build = Build(upload=BuildUpload(md5_checksum="a" * 32,
                                       request_id="r1"))
    assert c.object_built_image_url(m).endswith(":" + "a" * 32)


def test_artifact_url_hash():
    # bucket path is md5 of "clusters/{c}/namespaces/{ns}/{kind}s/{name}"
    # (reference internal/cloud/common.go:45-66)
    c = kind_cloud()
    m = Model(name="m1", namespace="ns1")
    inp = object_hash_input("kind", m)
    assert inp == "clusters/kind/namespaces/ns1/models/m1"
    url = c.object_artifact_url(m)
    assert str(url) == \
        f"tar:///bucket/{hashlib.md5(inp.encode()).hexdigest()}"


def test_gcp_autoconfigure():
    c = gcp_cloud()
    assert c.registry_url == "us-central1-docker.pkg.dev/proj/substratus"
    assert c.artifact_bucket_url.bucket == "proj-substratus-artifacts"
    assert c.artifact_bucket_url.scheme == "gs"


def test_bucket_url_parse():
    u = parse_bucket_url("gs://bkt/some/path")
    assert (u.scheme, u.bucket, u.path) == ("gs", "bkt", "some/path")
    u = parse_bucket_url("tar:///bucket")
    assert (u.scheme, u.bucket, u.path) == ("tar", "", "bucket")


@pytest.mark.parametrize("value,expect", [
    ("plain", {"name": "K", "value": "plain"}),
    ("${{ secrets.my-name.my-key }}",
     {"name": "K", "valueFrom": {"secretKeyRef": {"name": "my-name",
                                                  "key": "my-key"}}}),
    ("${{secrets.a.b}}",
     {"name": "K", "valueFrom": {"secretKeyRef": {"name": "a", "key": "b"}}}),
])
def test_resolve_env(value, expect):
    # (reference internal/controller/utils_test.go:11-30)
    assert resolve_env({"K": value}) == [expect]


def test_resources_apply_gpu():
    pod_meta, pod_spec = {}, {"containers": [{"name": "c"}]}
    r = Resources(cpu=4, memory=32, disk=100,
                  gpu=GPUResources(type="amd-mi355x", count=8))
    res.apply(pod_meta, pod_spec, "c", "gcp", r)
    c = pod_spec["containers"][0]
    assert c["resources"]["requests"]["amd.com/gpu"] == "8"
    assert c["resources"]["limits"]["amd.com/gpu"] == "8"
    assert c["resources"]["requests"]["memory"] == "32Gi"
    assert pod_spec["nodeSelector"][res.AMD_PRODUCT_LABEL] == "MI355X"


def test_resources_nvidia_types_map_to_mi355x():
    # reference example manifests name nvidia-l4/a100; they must schedule
    # onto the MI355X pool unchanged.
    for t in ("nvidia-l4", "nvidia-t4", "nvidia-a100"):
        info = res.get_gpu_info("gcp", t)
        assert info.resource_name == "amd.com/gpu"
        assert info.node_selector[res.AMD_PRODUCT_LABEL] == "MI355X"


def test_gcp_mount_bucket_annotations():
    # (reference internal/cloud/gcp_test.go: FUSE mount mutation)
    from runbooks_amd.cloud import Mount, MountBucketConfig
    c = gcp_cloud()
    m = Model(name="m1", namespace="ns1")
    meta, spec = {}, {"containers": [{"name": "serve"}]}
    c.mount_bucket(meta, spec, m, MountBucketConfig(
        name="model", container="serve",
        mounts=[Mount("artifacts", "model")], read_only=True))
    assert meta["annotations"]["gke-gcsfuse/volumes"] == "true"
    vol = spec["volumes"][0]
    assert vol["csi"]["driver"] == "gcsfuse.csi.storage.gke.io"
    vm = spec["containers"][0]["volumeMounts"][0]
    assert vm["mountPath"] == "/content/model"
    assert vm["readOnly"] is True


def test_manifest_roundtrip():
    d = {
        "apiVersion": "substratus.ai/v1", "kind": "Model",
        "metadata": {"name": "falcon-7b", "namespace": "default"},
        "spec": {
            "image": "img:1",
            "params": {"name": "falcon-7b", "steps": 100},
            "resources": {"gpu": {"type": "amd-mi355x", "count": 4}},
        },
    }
    m = object_from_manifest(d)
    assert isinstance(m, Model)
    assert m.params["steps"] == 100
    assert m.resources.gpu.count == 4
    back = m.to_dict()
    assert back["spec"]["params"] == d["spec"]["params"]
    assert object_from_manifest({"apiVersion": "v1", "kind": "Pod"}) is None


def test_all_yaml_manifests_parse():
    """Every committed YAML under config/, examples/, images/ parses and
    substratus manifests decode into typed objects."""
    import glob

    import yaml as _yaml
    files = (glob.glob("config/**/*.yaml", recursive=True) +
             glob.glob("examples/**/*.yaml", recursive=True))
    assert len(files) > 15
    n_sub = 0
    for f in files:
        with open(f) as fh:
            raw = fh.read()
        # examples use the ${{ secrets.* }} env syntax that is not YAML-safe
        import re
        raw = re.sub(r"\$\{\{[^}]*\}\}", "SECRETREF", raw)
        for doc in _yaml.safe_load_all(raw):
            if doc is None:
                continue
            assert isinstance(doc, dict), f
            obj = object_from_manifest(doc)
            if obj is not None:
                n_sub += 1
    assert n_sub >= 10  # the examples decode into typed objects


def test_crd_manifests_match_types():
    """config/crd/bases is in sync with the generator."""
    import yaml as _yaml

    from runbooks_amd.api.crd import crd_manifest
    from runbooks_amd.api.types import KINDS, PLURALS
    for kind in KINDS:
        path = f"config/crd/bases/substratus.ai_{PLURALS[kind]}.yaml"
        with open(path) as f:
            on_disk = _yaml.safe_load(f)
        assert on_disk == crd_manifest(kind), f"stale {path}; run make manifests"


@pytest.mark.skipif(not os.path.isdir("/root/reference/examples"),
                    reason="reference checkout not present")
def test_reference_example_manifests_apply_unchanged():
    """Field compatibility, proven against the REFERENCE's own example
    manifests: every substratus.ai/v1 document under the reference's
    examples/ decodes into our typed objects and round-trips its spec
    (the 'manifests apply unchanged' claim)."""
    import glob
    import re

    import yaml as _yaml

    files = glob.glob("/root/reference/examples/**/*.yaml", recursive=True)
    assert len(files) >= 10
    n = 0
    for f in files:
        raw = re.sub(r"\$\{\{[^}]*\}\}", "SECRETREF", open(f).read())
        for doc in _yaml.safe_load_all(raw):
            if not doc or doc.get("apiVersion") != "substratus.ai/v1":
                continue
            obj = object_from_manifest(doc)
            assert obj is not None, f
            back = obj.to_dict()
            assert back["kind"] == doc["kind"], f
            assert back["metadata"]["name"] == doc["metadata"]["name"]
            spec = doc.get("spec", {})
            bspec = back.get("spec", {})
            # every field the reference manifest sets survives round-trip
            # (modelName/datasetName: legacy flat aliases normalized to
            # the CRD's ObjectRef form)
            alias = {"modelName": "model", "datasetName": "dataset"}
            for key, val in spec.items():
                if key in alias:
                    assert bspec[alias[key]]["name"] == val, (f, key)
                    continue
                assert key in bspec, (f, key)
                if isinstance(val, (str, int, bool)):
                    assert bspec[key] == val, (f, key, bspec[key], val)
            n += 1
    assert n >= 15, n
