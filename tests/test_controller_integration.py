"""Integration tests: reconcilers against the in-memory API server.

Mirrors the reference envtest suite (reference
internal/controller/main_test.go:46-191): all reconcilers run against a
fake cloud (GCP with hardcoded ids) and the fake SCI client; no pods
actually run, so tests fake Job/Pod/Deployment status exactly like
fakeJobComplete / fakePodReady (reference main_test.go:245-265).
"""
import hashlib

import pytest

from runbooks_amd.api import Build, BuildGit, BuildUpload
from runbooks_amd.api.types import (
    Dataset,
    GPUResources,
    Model,
    Notebook,
    ObjectRef,
    Resources,
    Server,
)
from runbooks_amd.cloud import new_cloud
from runbooks_amd.controller import ControllerManager
from runbooks_amd.k8s import MemoryKubeClient
from runbooks_amd.sci import FakeSCIClient

API = "substratus.ai/v1"


@pytest.fixture()
def env():
    kube = MemoryKubeClient()
    cloud = new_cloud({
        "CLOUD": "gcp", "CLUSTER_NAME": "test", "PROJECT_ID": "test-project",
        "CLUSTER_LOCATION": "us-central1-a",
        "PRINCIPAL": "substratus@test-project.iam.gserviceaccount.com"})
    sci = FakeSCIClient()
    mgr = ControllerManager(kube, cloud, sci)
    return kube, cloud, sci, mgr


def fake_job_complete(kube, ns, name):
    """(reference main_test.go:245-255)"""
    kube.patch("batch/v1", "Job", ns, name, {"status": {
        "succeeded": 1,
        "conditions": [{"type": "Complete", "status": "True"}]}})


def fake_job_failed(kube, ns, name):
    kube.patch("batch/v1", "Job", ns, name, {"status": {
        "conditions": [{"type": "Failed", "status": "True"}]}})


def fake_pod_ready(kube, ns, name):
    """(reference main_test.go:257-265)"""
    kube.patch("v1", "Pod", ns, name, {"status": {
        "phase": "Running",
        "conditions": [{"type": "Ready", "status": "True"}]}})


def fake_deployment_ready(kube, ns, name):
    kube.patch("apps/v1", "Deployment", ns, name,
               {"status": {"readyReplicas": 1}})


def get_model(kube, name, ns="default"):
    return Model.from_dict(kube.get(API, "Model", ns, name))


# ---------------------------------------------------------------------------
# Build flows (reference model_controller_test.go:20-80, testContainerBuild)
# ---------------------------------------------------------------------------

def test_git_build_flow(env):
    kube, cloud, sci, mgr = env
    m = Model(name="m-git", image=None,
              build=Build(git=BuildGit(url="https://github.com/x/y",
                                       branch="main")))
    kube.create(m.to_dict())
    mgr.reconcile_all()

    # builder SA annotated with the workload-identity principal
    sa = kube.get("v1", "ServiceAccount", "default", "container-builder")
    assert sa["metadata"]["annotations"]["iam.gke.io/gcp-service-account"] \
        == cloud.principal

    # -bld job created with the kaniko shape
    job = kube.get("batch/v1", "Job", "default", "m-git-model-bld")
    assert job is not None
    podspec = job["spec"]["template"]["spec"]
    assert podspec["initContainers"][0]["image"] == "alpine/git"
    assert "--branch" in podspec["initContainers"][0]["args"]
    assert podspec["containers"][0]["name"] == "builder"
    assert job["metadata"]["annotations"]["image"] == \
        cloud.object_built_image_url(m)

    got = get_model(kube, "m-git")
    c = got.get_condition("Built")
    assert c and c["status"] == "False" and c["reason"] == "JobNotComplete"

    fake_job_complete(kube, "default", "m-git-model-bld")
    mgr.reconcile_all()
    got = get_model(kube, "m-git")
    assert got.is_condition_true("Built")
    assert got.get_image() == cloud.object_built_image_url(got)


def test_upload_build_handshake(env):
    kube, cloud, sci, mgr = env
    md5 = "a" * 32
    sci.signed_url = "https://signed.example/put"
    m = Model(name="m-up",
              build=Build(upload=BuildUpload(md5_checksum=md5,
                                             request_id="req-1")))
    kube.create(m.to_dict())
    mgr.reconcile_all(rounds=1)

    # handshake step 1: signed URL in status, Uploaded=False AwaitingUpload
    got = get_model(kube, "m-up")
    assert got.build_upload.signed_url == "https://signed.example/put"
    assert got.build_upload.request_id == "req-1"
    c = got.get_condition("Uploaded")
    assert c["status"] == "False" and c["reason"] == "AwaitingUpload"
    # SCI was asked for a URL for uploads/latest.tar.gz under the hash path
    call = [c for c in sci.calls if c[0] == "CreateSignedURL"][0]
    assert call[2].endswith("uploads/latest.tar.gz")

    # step 2: client uploaded; SCI now reports the matching md5
    sci.object_md5 = md5
    mgr.reconcile_all(rounds=1)
    got = get_model(kube, "m-up")
    assert got.is_condition_true("Uploaded")
    assert got.build_upload.stored_md5_checksum == md5
    assert got.build_upload.signed_url == ""

    # kaniko storage-context job exists; completing it sets the image
    fake_job_complete(kube, "default", "m-up-model-bld")
    mgr.reconcile_all()
    got = get_model(kube, "m-up")
    assert got.is_condition_true("Built")
    assert got.get_image().endswith(":" + md5)


def test_upload_reuses_existing_storage_object(env):
    # edge case reference build_reconciler.go:192-210
    kube, cloud, sci, mgr = env
    md5 = "b" * 32
    sci.object_md5 = md5  # already in storage
    m = Model(name="m-reuse",
              build=Build(upload=BuildUpload(md5_checksum=md5,
                                             request_id="req-9")))
    kube.create(m.to_dict())
    mgr.reconcile_all(rounds=1)
    got = get_model(kube, "m-reuse")
    c = got.get_condition("Uploaded")
    assert c["status"] == "True" and c["reason"] == "UploadFound"
    # no signed URL round-trip was needed
    assert got.build_upload.signed_url == ""


# ---------------------------------------------------------------------------
# Model flows (reference model_controller_test.go:80-159)
# ---------------------------------------------------------------------------

def test_model_import_job_and_params(env):
    kube, cloud, sci, mgr = env
    m = Model(name="imp", image="img:1",
              params={"name": "facebook/opt-125m"})
    kube.create(m.to_dict())
    mgr.reconcile_all(rounds=1)

    # params ConfigMap (reference testParamsConfigMap)
    cm = kube.get("v1", "ConfigMap", "default", "imp-model-params")
    assert "facebook/opt-125m" in cm["data"]["params.json"]

    job = kube.get("batch/v1", "Job", "default", "imp-modeller")
    podspec = job["spec"]["template"]["spec"]
    c = podspec["containers"][0]
    assert c["image"] == "img:1"
    vm = {v["mountPath"] for v in c["volumeMounts"]}
    assert "/content/params.json" in vm
    assert "/content/artifacts" in vm

    got = get_model(kube, "imp")
    assert not got.ready
    fake_job_complete(kube, "default", "imp-modeller")
    mgr.reconcile_all(rounds=1)
    got = get_model(kube, "imp")
    assert got.ready and got.is_condition_true("Complete")
    assert got.artifacts.url.startswith("gs://test-project-substratus")


def test_model_trainer_chain_gating(env):
    kube, cloud, sci, mgr = env
    base = Model(name="base", image="img:base")
    ds = Dataset(name="squad", image="img:ds")
    trained = Model(name="trained", image="img:train",
                    model=ObjectRef("base"), dataset=ObjectRef("squad"),
                    resources=Resources(gpu=GPUResources(type="amd-mi355x",
                                                         count=8)))
    kube.create(trained.to_dict())
    mgr.reconcile_all(rounds=1)
    got = get_model(kube, "trained")
    assert got.get_condition("Complete")["reason"] == "BaseModelNotFound"

    kube.create(base.to_dict())
    kube.create(ds.to_dict())
    mgr.reconcile_all(rounds=1)
    got = get_model(kube, "trained")
    assert got.get_condition("Complete")["reason"] in (
        "BaseModelNotReady", "ReasonDatasetNotReady")

    fake_job_complete(kube, "default", "base-modeller")
    fake_job_complete(kube, "default", "squad-data-loader")
    mgr.reconcile_all()
    # trainer job now exists with dataset+model RO mounts and GPU resources
    job = kube.get("batch/v1", "Job", "default", "trained-modeller")
    podspec = job["spec"]["template"]["spec"]
    c = podspec["containers"][0]
    mounts = {v["mountPath"]: v for v in c["volumeMounts"]}
    assert mounts["/content/data"]["readOnly"] is True
    assert mounts["/content/model"]["readOnly"] is True
    assert mounts["/content/artifacts"].get("readOnly", False) is False
    assert c["resources"]["limits"]["amd.com/gpu"] == "8"
    # GPU jobs don't retry (reference model_controller.go:294-303)
    assert job["spec"]["backoffLimit"] == 0

    fake_job_complete(kube, "default", "trained-modeller")
    mgr.reconcile_all(rounds=1)
    assert get_model(kube, "trained").ready


def test_model_job_failure_surfaces(env):
    kube, cloud, sci, mgr = env
    kube.create(Model(name="bad", image="img:1").to_dict())
    mgr.reconcile_all(rounds=1)
    fake_job_failed(kube, "default", "bad-modeller")
    mgr.reconcile_all(rounds=1)
    got = get_model(kube, "bad")
    assert not got.ready
    assert got.get_condition("Complete")["reason"] == "JobFailed"


# ---------------------------------------------------------------------------
# Server flow (reference server_controller_test.go:17-77)
# ---------------------------------------------------------------------------

def test_server_flow(env):
    kube, cloud, sci, mgr = env
    kube.create(Model(name="m7b", image="img:m").to_dict())
    kube.create(Server(name="srv", image="img:srv",
                       model=ObjectRef("m7b"),
                       command=["serve", "--tp", "1"]).to_dict())
    mgr.reconcile_all(rounds=1)
    srv = Server.from_dict(kube.get(API, "Server", "default", "srv"))
    assert srv.get_condition("Serving")["reason"] == "ModelNotReady"

    fake_job_complete(kube, "default", "m7b-modeller")
    mgr.reconcile_all()

    svc = kube.get("v1", "Service", "default", "srv-server")
    port = svc["spec"]["ports"][0]
    assert port["port"] == 8080 and port["targetPort"] == "http-serve"
    dep = kube.get("apps/v1", "Deployment", "default", "srv-server")
    c = dep["spec"]["template"]["spec"]["containers"][0]
    assert c["command"] == ["serve", "--tp", "1"]
    assert c["readinessProbe"]["httpGet"]["path"] == "/"
    assert {v["mountPath"] for v in c["volumeMounts"]} >= {
        "/content/model", "/content/params.json"}

    srv = Server.from_dict(kube.get(API, "Server", "default", "srv"))
    assert not srv.ready
    fake_deployment_ready(kube, "default", "srv-server")
    mgr.reconcile_all(rounds=1)
    srv = Server.from_dict(kube.get(API, "Server", "default", "srv"))
    assert srv.ready and srv.is_condition_true("Serving")


# ---------------------------------------------------------------------------
# Notebook flow (reference notebook_controller_test.go:20-85)
# ---------------------------------------------------------------------------

def test_notebook_flow_and_suspend(env):
    kube, cloud, sci, mgr = env
    kube.create(Notebook(name="nb", image="img:nb").to_dict())
    mgr.reconcile_all(rounds=1)

    pod = kube.get("v1", "Pod", "default", "nb-notebook")
    c = pod["spec"]["containers"][0]
    assert c["command"][:2] == ["jupyter", "lab"]
    assert "--NotebookApp.token=$(NOTEBOOK_TOKEN)" in c["command"]
    assert c["ports"][0]["containerPort"] == 8888
    assert c["readinessProbe"]["httpGet"]["path"] == "/api"

    fake_pod_ready(kube, "default", "nb-notebook")
    mgr.reconcile_all(rounds=1)
    nb = Notebook.from_dict(kube.get(API, "Notebook", "default", "nb"))
    assert nb.ready and nb.get_condition("Serving")["reason"] == "PodReady"

    # suspend deletes the pod (reference notebook_controller.go:134-155)
    nb.suspend = True
    kube.apply(nb.to_dict())
    mgr.reconcile_all(rounds=1)
    assert kube.get("v1", "Pod", "default", "nb-notebook") is None
    nb = Notebook.from_dict(kube.get(API, "Notebook", "default", "nb"))
    assert not nb.ready
    assert nb.get_condition("Serving")["reason"] == "Suspended"


def test_leader_election(env):
    """Lease-based leader election: first holder wins, a second identity
    waits until the lease goes stale, same identity renews."""
    kube, cloud, sci, mgr = env
    from runbooks_amd.controller.manager import LeaderElector

    a = LeaderElector(kube, "mgr-a", lease_seconds=15)
    b = LeaderElector(kube, "mgr-b", lease_seconds=15)
    assert a.try_acquire()
    assert not b.try_acquire()
    assert a.try_acquire()          # renewal by the holder
    # stale lease (old renewTime) is taken over
    kube.patch("coordination.k8s.io/v1", "Lease", "substratus",
               a.name, {"spec": {"renewTime": "2000-01-01T00:00:00.000Z"}})
    assert b.try_acquire()


def test_manager_watch_loop(env):
    """The production watch-driven loop: events in the in-memory API
    server flow through watches -> workqueue -> reconcilers, including
    the dependency fan-out (Model ready -> dependent Server reconciled)."""
    import threading
    import time

    kube, cloud, sci, mgr = env
    t = threading.Thread(target=mgr.run, daemon=True)
    t.start()
    try:
        kube.create(Model(name="wm", image="img:1").to_dict())

        def wait_for(fn, timeout=10.0):
            end = time.time() + timeout
            while time.time() < end:
                if fn():
                    return True
                time.sleep(0.02)
            return False

        assert wait_for(lambda: kube.get("batch/v1", "Job", "default",
                                         "wm-modeller") is not None)
        kube.create(Server(name="ws", image="img:s",
                           model=ObjectRef("wm")).to_dict())
        assert wait_for(lambda: (kube.get(API, "Server", "default", "ws")
                                 .get("status") or {}).get("conditions"))
        # completing the Job flips the Model; the watch fan-out then
        # reconciles the Server into creating its Deployment
        fake_job_complete(kube, "default", "wm-modeller")
        assert wait_for(lambda: kube.get("apps/v1", "Deployment", "default",
                                         "ws-server") is not None)
    finally:
        mgr.stop()
        t.join(timeout=3)


def test_manager_metrics(env):
    """Reconcile counters/latency exported for Prometheus (the reference
    exposes controller-runtime metrics behind kube-rbac-proxy)."""
    kube, cloud, sci, mgr = env
    from prometheus_client import REGISTRY

    kube.create(Model(name="metr", image="img:1").to_dict())
    mgr.reconcile_object("Model", "default", "metr")
    v = REGISTRY.get_sample_value("rb_reconciles_total",
                                  {"kind": "Model"})
    assert v and v >= 1


def test_build_job_recreated_when_out_of_date(env):
    """Changing the build spec (new branch -> new image URL) deletes the
    stale kaniko Job and creates a fresh one
    (reference build_reconciler.go:128-136)."""
    kube, cloud, sci, mgr = env
    m = Model(name="m-ood",
              build=Build(git=BuildGit(url="https://x", branch="v1")))
    kube.create(m.to_dict())
    mgr.reconcile_all(rounds=1)
    job1 = kube.get("batch/v1", "Job", "default", "m-ood-model-bld")
    assert job1["metadata"]["annotations"]["image"].endswith(":v1")

    kube.patch("substratus.ai/v1", "Model", "default", "m-ood",
               {"spec": {"build": {"git": {"url": "https://x",
                                           "branch": "v2"}}}})
    mgr.reconcile_all(rounds=1)
    job2 = kube.get("batch/v1", "Job", "default", "m-ood-model-bld")
    assert job2["metadata"]["annotations"]["image"].endswith(":v2")
    # fresh object (recreated, not patched)
    assert job2["metadata"]["uid"] != job1["metadata"]["uid"]


@pytest.mark.parametrize("seed", [7, 42, 123])
def test_controller_fuzz_random_object_graph(env, seed):
    """Random Models/Datasets/Servers/Notebooks with random (possibly
    dangling) references: reconciliation must never wedge or corrupt —
    every object ends with a coherent status once its chain is unblocked."""
    import random

    kube, cloud, sci, mgr = env
    rng = random.Random(seed)
    names = [f"o{i}" for i in range(12)]
    for i, n in enumerate(names):
        kind = rng.choice(["Model", "Dataset", "Server", "Notebook"])
        ref = rng.choice(names + ["dangling"])
        if kind == "Model":
            o = Model(name=n, image="img:x",
                      model=ObjectRef(ref) if rng.random() < 0.4 else None,
                      dataset=ObjectRef(ref) if rng.random() < 0.3 else None)
        elif kind == "Dataset":
            o = Dataset(name=n, image="img:x")
        elif kind == "Server":
            o = Server(name=n, image="img:x", model=ObjectRef(ref))
        else:
            o = Notebook(name=n, image="img:x",
                         model=ObjectRef(ref) if rng.random() < 0.5 else None)
        kube.create(o.to_dict())

    for _ in range(4):
        mgr.reconcile_all(rounds=2)
        # complete any Jobs that appeared, unblocking chains
        for j in kube.list("batch/v1", "Job"):
            fake_job_complete(kube, "default", j["metadata"]["name"])

    # invariant: every object has a status and either ready or a
    # condition explaining why not
    for kind in ("Model", "Dataset", "Server", "Notebook"):
        for raw in kube.list(API, kind):
            st = raw.get("status") or {}
            assert "ready" in st, raw["metadata"]["name"]
            if not st["ready"]:
                assert st.get("conditions"), \
                    f"{kind}/{raw['metadata']['name']} stuck with no reason"


def test_manager_watch_disconnect_relists(env):
    """A dropped watch must not lose events: objects created while the
    watch is down get picked up by the re-list on reconnection
    (controller-runtime informer semantics)."""
    import threading
    import time

    kube, cloud, sci, mgr = env

    # wrap the client's watch: the FIRST stream dies mid-flight, the
    # second one works — simulating an apiserver disconnect
    orig_watch = kube.watch
    fail_once = {"Model": True}

    def flaky_watch(api_version, kind, namespace="", stop=None):
        if fail_once.get(kind):
            fail_once[kind] = False
            raise ConnectionResetError("apiserver went away")
        return orig_watch(api_version, kind, namespace=namespace, stop=stop)

    kube.watch = flaky_watch
    # object created while the Model watch is down
    kube.create(Model(name="lost-ev", image="img:1").to_dict())

    t = threading.Thread(target=mgr.run, daemon=True)
    t.start()
    try:
        end = time.time() + 10
        while time.time() < end:
            if kube.get("batch/v1", "Job", "default",
                        "lost-ev-modeller") is not None:
                break
            time.sleep(0.02)
        assert kube.get("batch/v1", "Job", "default",
                        "lost-ev-modeller") is not None, \
            "re-list after watch failure did not reconcile the object"
    finally:
        mgr.stop()
        t.join(timeout=3)
        kube.watch = orig_watch
