"""Client library + CLI tests, including the full upload handshake
"system test": CLI client <-> controllers <-> real kind SCI (gRPC + HTTP)
against the in-memory API server — the offline analog of the reference's
test/system.sh kind flow.
"""
import os
import threading

import pytest

from runbooks_amd import client as sclient
from runbooks_amd.api.types import Dataset, Model, Notebook, ObjectRef, Server
from runbooks_amd.cloud import new_cloud
from runbooks_amd.controller import ControllerManager
from runbooks_amd.k8s import MemoryKubeClient
from runbooks_amd.sci import ControllerClient
from runbooks_amd.sci.kind_server import KindSCI, make_http_server
from runbooks_amd.sci.server import serve as sci_serve


def test_prepare_tarball_requires_dockerfile(tmp_path):
    with pytest.raises(FileNotFoundError):
        sclient.prepare_image_tarball(str(tmp_path))
    (tmp_path / "Dockerfile").write_text("FROM scratch\n")
    (tmp_path / "train.py").write_text("print('hi')\n")
    tb = sclient.prepare_image_tarball(str(tmp_path))
    assert len(tb.md5_checksum) == 32
    assert os.path.exists(tb.path)


def test_notebook_for_object_conversions():
    m = Model(name="m", image="img:1", model=ObjectRef("base"),
              dataset=ObjectRef("d"), params={"x": 1})
    nb = sclient.notebook_for_object(m)
    assert isinstance(nb, Notebook)
    assert nb.name == "m-model" and nb.model.name == "base"
    s = Server(name="s", image="img:2", model=ObjectRef("m"))
    nb2 = sclient.notebook_for_object(s)
    assert nb2.name == "s-server" and nb2.model.name == "m"
    d = Dataset(name="d", image="img:3")
    assert sclient.notebook_for_object(d).name == "d-dataset"


def test_upload_handshake_system(tmp_path):
    """End-to-end: tarball -> apply -> controller signed URL (real kind SCI
    over gRPC) -> HTTP PUT -> controller verifies md5 -> Built image."""
    kube = MemoryKubeClient()
    cloud = new_cloud({"CLOUD": "kind", "CLUSTER_NAME": "kind",
                       "REGISTRY_PORT_5000_TCP_ADDR": "10.0.0.9"})

    sci_impl = KindSCI(root=str(tmp_path))
    httpd = make_http_server(sci_impl, port=0)
    sci_impl.signed_url_address = f"http://127.0.0.1:{httpd.server_address[1]}"
    grpc_server = sci_serve(sci_impl, "127.0.0.1:0")
    sci_client = ControllerClient(f"127.0.0.1:{grpc_server.bound_port}")

    mgr = ControllerManager(kube, cloud, sci_client)

    build_dir = tmp_path / "ctx"
    build_dir.mkdir()
    (build_dir / "Dockerfile").write_text("FROM scratch\nCOPY . /src\n")
    (build_dir / "main.py").write_text("print('model')\n")

    obj = Model(name="up1")
    tb = sclient.prepare_image_tarball(str(build_dir))
    sclient.set_upload_container_spec(obj, tb, request_id="req-abc")
    kube.apply(obj.to_dict())

    # background reconciler thread stands in for the running operator
    stop = threading.Event()

    def loop():
        while not stop.is_set():
            mgr.reconcile_all(rounds=1)
            stop.wait(0.05)

    t = threading.Thread(target=loop, daemon=True)
    t.start()
    try:
        sclient.upload(kube, obj, tb, timeout=20)
        # controller verifies the stored md5 and marks Uploaded
        deadline = threading.Event()
        for _ in range(100):
            raw = kube.get("substratus.ai/v1", "Model", "default", "up1")
            got = Model.from_dict(raw)
            if got.is_condition_true("Uploaded"):
                break
            deadline.wait(0.05)
        assert got.is_condition_true("Uploaded")
        assert got.build_upload.stored_md5_checksum == tb.md5_checksum
        # kaniko storage job exists; completing it publishes the image
        kube.patch("batch/v1", "Job", "default", "up1-model-bld", {"status": {
            "succeeded": 1, "conditions": [{"type": "Complete",
                                            "status": "True"}]}})
        for _ in range(100):
            raw = kube.get("substratus.ai/v1", "Model", "default", "up1")
            got = Model.from_dict(raw)
            if got.get_image():
                break
            deadline.wait(0.05)
        assert got.get_image().endswith(":" + tb.md5_checksum)
    finally:
        stop.set()
        t.join(timeout=2)
        grpc_server.stop(0)
        httpd.shutdown()


def test_cli_manifest_discovery(tmp_path):
    from runbooks_amd.cli.manifests import find_manifests, next_version_name
    (tmp_path / "model.yaml").write_text(
        "apiVersion: substratus.ai/v1\nkind: Model\n"
        "metadata: {name: m1}\nspec: {image: i}\n---\n"
        "apiVersion: v1\nkind: ConfigMap\nmetadata: {name: x}\n")
    (tmp_path / "server.yaml").write_text(
        "apiVersion: substratus.ai/v1\nkind: Server\n"
        "metadata: {name: s1}\nspec: {image: i, model: {name: m1}}\n")
    objs = find_manifests(str(tmp_path))
    assert [o.kind for o in objs] == ["Model", "Server"]
    assert find_manifests(str(tmp_path), kind_filter="server")[0].name == "s1"

    kube = MemoryKubeClient()
    assert next_version_name(kube, "Model", "default", "m") == "m"
    kube.create(Model(name="m").to_dict())
    assert next_version_name(kube, "Model", "default", "m") == "m-1"
    kube.create(Model(name="m-4").to_dict())
    assert next_version_name(kube, "Model", "default", "m") == "m-5"
    assert next_version_name(kube, "Model", "default", "m-2") == "m-5"


def test_cli_get_delete(monkeypatch, capsys):
    import importlib
    from click.testing import CliRunner
    cli_main = importlib.import_module("runbooks_amd.cli.main")

    kube = MemoryKubeClient()
    kube.create(Model(name="m1", image="i").to_dict())
    monkeypatch.setattr(cli_main, "_kube", lambda: kube)
    r = CliRunner().invoke(cli_main.main, ["get", "models"])
    assert r.exit_code == 0 and "m1" in r.output
    r = CliRunner().invoke(cli_main.main, ["delete", "model", "m1"])
    assert r.exit_code == 0
    assert kube.get("substratus.ai/v1", "Model", "default", "m1") is None


def test_nbwatch_events(tmp_path):
    from runbooks_amd import nbwatch
    root = tmp_path / "content"
    (root / "data").mkdir(parents=True)   # special dir: ignored
    (root / "nb").mkdir()
    f = root / "nb" / "train.py"
    gen = nbwatch.watch(str(root), interval=0.01, once=True)
    f.write_text("x = 1\n")
    (root / "data" / "ignored.bin").write_text("z")
    events = list(gen)
    assert {(e["op"], os.path.basename(e["path"])) for e in events} == {
        ("CREATE", "train.py")}


def test_entrypoint_params_to_env(tmp_path, monkeypatch):
    """The contract's params.json -> PARAM_* conversion
    (docs/container-contract.md; reference container-contract.md:34-48)."""
    import json
    from runbooks_amd.workloads.entrypoint import params_to_env
    assert params_to_env({"epochs": 1, "use-lora": True, "name": "x/y"}) == {
        "PARAM_EPOCHS": "1", "PARAM_USE_LORA": "true", "PARAM_NAME": "x/y"}


def test_dataset_loader_synthetic(tmp_path, monkeypatch):
    from runbooks_amd.workloads import dataset_loader
    monkeypatch.setenv("ARTIFACTS_DIR", str(tmp_path))
    monkeypatch.setenv("PARAM_SYNTHETIC", "true")
    assert dataset_loader.main() == 0
    lines = (tmp_path / "data.jsonl").read_text().strip().splitlines()
    assert len(lines) == 256


def test_model_loader_synthetic(tmp_path, monkeypatch):
    from runbooks_amd.workloads import model_loader
    monkeypatch.setenv("ARTIFACTS_DIR", str(tmp_path))
    monkeypatch.setenv("PARAM_NAME", "tiny-llama")
    monkeypatch.setenv("PARAM_SYNTHETIC", "true")
    assert model_loader.main() == 0
    assert (tmp_path / "model.safetensors").exists()
    assert "tiny-llama" in (tmp_path / "config.json").read_text()


def test_trainer_main_end_to_end(tmp_path, monkeypatch):
    """The trainer image main on CPU: loader artifacts -> fine-tune ->
    checkpoints in /content/artifacts (contract round trip)."""
    from runbooks_amd.workloads import dataset_loader, model_loader, trainer_main
    model_dir = tmp_path / "model"
    data_dir = tmp_path / "data"
    out_dir = tmp_path / "artifacts"
    model_dir.mkdir(), data_dir.mkdir()
    monkeypatch.setenv("ARTIFACTS_DIR", str(model_dir))
    monkeypatch.setenv("PARAM_NAME", "tiny-llama")
    monkeypatch.setenv("PARAM_SYNTHETIC", "true")
    assert model_loader.main() == 0
    monkeypatch.setenv("ARTIFACTS_DIR", str(data_dir))
    assert dataset_loader.main() == 0

    monkeypatch.setenv("MODEL_DIR", str(model_dir))
    monkeypatch.setenv("DATA_DIR", str(data_dir))
    monkeypatch.setenv("ARTIFACTS_DIR", str(out_dir))
    monkeypatch.setenv("PARAM_NUM_TRAIN_STEPS", "2")
    monkeypatch.setenv("PARAM_SAVE_STEPS", "2")
    monkeypatch.setenv("PARAM_SEQ_LEN", "32")
    monkeypatch.setenv("PARAM_PER_DEVICE_TRAIN_BATCH_SIZE", "2")
    monkeypatch.delenv("PARAM_SYNTHETIC")
    assert trainer_main.main() == 0
    ckpts = list(out_dir.glob("checkpoint-*"))
    assert ckpts, "trainer wrote no checkpoint"
    assert (ckpts[0] / "model.safetensors").exists()
    # artifact-completeness markers (reference design.md "Buckets")
    import json as _json
    for d in (model_dir, data_dir, out_dir):
        assert _json.loads((d / "completed.json").read_text())["completed"]


def test_bench_contract_cpu(tmp_path):
    """bench.py default emits BOTH driver-contract JSON lines (finetune
    first, serve last — BASELINE.json names both headline metrics)."""
    import json as _json
    import subprocess
    import sys

    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--micro-batch", "2", "--seq-len", "16"],
        capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-800:]
    lines = [ln for ln in out.stdout.strip().splitlines() if ln.startswith("{")]
    assert len(lines) == 2
    d = _json.loads(lines[0])
    assert d["metric"] == "finetune_samples_per_sec"
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["scaling"] == "weak"
    assert set(d) >= {"value", "unit", "ms_per_step", "higher_is_better",
                      "vs_baseline", "dtype", "data", "config"}
    d2 = _json.loads(lines[1])
    assert d2["metric"] == "serve_tokens_per_sec"
    assert d2["config"]["parallelism"] == "tp1"

    out = subprocess.run(
        [sys.executable, "bench.py", "--mode", "serve", "--steps", "3",
         "--warmup", "1", "--serve-batch", "2", "--prompt-len", "8"],
        capture_output=True, text=True, timeout=240)
    assert out.returncode == 0, out.stderr[-800:]
    d = _json.loads(out.stdout.strip().splitlines()[-1])
    assert d["metric"] == "serve_tokens_per_sec"
    assert d["value"] > 0


def test_tui_components():
    from runbooks_amd.tui import format_conditions, select_manifest

    raw = {"status": {"ready": False, "conditions": [
        {"type": "Built", "status": "True", "reason": "JobComplete"},
        {"type": "Complete", "status": "False", "reason": "JobNotComplete"},
    ]}}
    rows = format_conditions(raw)
    assert rows == [("✓", "Built", "JobComplete"),
                    ("…", "Complete", "JobNotComplete"),
                    ("…", "Ready", "")]
    raw["status"]["ready"] = True
    assert format_conditions(raw)[-1] == ("✓", "Ready", "")

    m1, m2 = Model(name="a"), Model(name="b")
    # non-TTY path picks the first deterministically
    assert select_manifest([m1, m2]) is m1
    assert select_manifest([m2]) is m2


def test_cli_suspend_resume(monkeypatch):
    import importlib
    from click.testing import CliRunner
    cli_main = importlib.import_module("runbooks_amd.cli.main")

    kube = MemoryKubeClient()
    kube.create(Notebook(name="nb1", image="i").to_dict())
    monkeypatch.setattr(cli_main, "_kube", lambda: kube)
    r = CliRunner().invoke(cli_main.main, ["suspend", "nb1"])
    assert r.exit_code == 0
    raw = kube.get("substratus.ai/v1", "Notebook", "default", "nb1")
    assert raw["spec"]["suspend"] is True
    r = CliRunner().invoke(cli_main.main, ["resume", "nb1"])
    assert r.exit_code == 0
    raw = kube.get("substratus.ai/v1", "Notebook", "default", "nb1")
    assert raw["spec"]["suspend"] is False


def test_server_main_builds_engine(tmp_path, monkeypatch):
    """The server image main on CPU: config marker -> engine + tokenizer,
    handed to serve_forever (patched out; binding :8080 is the only part
    skipped). Covers the contract env parsing incl. MODEL_LOAD_IN_8BIT."""
    import json as _json
    from runbooks_amd.workloads import server_main
    model_dir = tmp_path / "model"
    model_dir.mkdir()
    (model_dir / "config.json").write_text(
        _json.dumps({"runbooks_amd_config": "tiny-llama"}))
    monkeypatch.setenv("MODEL_DIR", str(model_dir))
    monkeypatch.setenv("MODEL_LOAD_IN_8BIT", "true")  # bf16 fallback on CPU
    monkeypatch.delenv("TP", raising=False)

    got = {}

    def fake_serve(engine, tok, port, model_name):
        got.update(engine=engine, tok=tok, port=port, model=model_name)

    import runbooks_amd.serve.http as http_mod
    monkeypatch.setattr(http_mod, "serve_forever", fake_serve)
    assert server_main.main() == 0
    assert got["model"] == "tiny-llama" and got["port"] == 8080
    assert got["engine"].cfg.name == "tiny-llama"
    out = got["engine"].generate([1, 2, 3], max_new_tokens=2)
    assert len(out) == 2


def test_kubectl_plugin_shims(monkeypatch, capsys):
    """kubectl-notebook / kubectl-applybuild entry points forward to the
    sub subcommands (kubectl plugin discovery contract)."""
    import sys

    import pytest as _pytest

    import importlib
    cli_main = importlib.import_module("runbooks_amd.cli.main")

    for fn in (cli_main.kubectl_notebook, cli_main.kubectl_applybuild):
        monkeypatch.setattr(sys, "argv", ["x", "--help"])
        with _pytest.raises(SystemExit) as e:
            fn()
        assert e.value.code == 0
    out = capsys.readouterr().out
    assert "kubectl applybuild" in out
