"""`sub` command implementations.

Parity with the reference command surface (internal/cli/root.go:9-24):
  sub apply -f PATH        server-side apply manifests (+ upload builds)
  sub run PATH [-i|-r]     upload build context, create Model, wait ready
  sub notebook PATH        dev loop: notebook from manifest, sync, forward
  sub serve PATH           apply a Server and wait for readiness
  sub get KIND [NAME]      list/get substratus objects
  sub delete KIND NAME     delete an object
"""
from __future__ import annotations

import sys
import threading
import uuid

import click

from ..api.types import KINDS
from ..k8s import HTTPKubeClient
from .. import client as sclient
from .manifests import find_manifests, next_version_name

try:
    from rich.console import Console
    _console = Console()

    def _say(msg):
        _console.print(msg)
except ImportError:  # rich is in the image, but degrade gracefully
    def _say(msg):
        print(msg)


def _kube():
    return HTTPKubeClient()


def _report_conditions(raw):
    conds = (raw.get("status") or {}).get("conditions") or []
    for c in conds:
        mark = "[green]✓[/green]" if c["status"] == "True" else "[yellow]…[/yellow]"
        _say(f"  {mark} {c['type']}: {c.get('reason', '')}")


def _apply_with_optional_upload(kube, obj, build_path=None, wait=False):
    from ..tui import ReadinessChecklist, UploadProgress
    if build_path is not None:
        with UploadProgress(f"{obj.kind}/{obj.name}") as up:
            tb = sclient.prepare_image_tarball(build_path,
                                               progress=up.on_file)
            sclient.set_upload_container_spec(obj, tb, uuid.uuid4().hex)
            sclient.clear_image(obj)
            kube.apply(obj.to_dict())
            _say(f"[bold]{obj.kind}/{obj.name}[/bold]: uploading build "
                 f"context (md5 {tb.md5_checksum[:12]}…)")
            sclient.upload(kube, obj, tb, progress=up.on_fraction)
    else:
        kube.apply(obj.to_dict())
        _say(f"[bold]{obj.kind}/{obj.name}[/bold]: applied")
    if wait:
        with ReadinessChecklist(f"{obj.kind}/{obj.name}") as view:
            raw = sclient.wait_ready(kube, obj, callback=view.update)
        _report_conditions(raw)


@click.group()
def main():
    """Substratus-compatible CLI for the MI355X-native platform."""


@main.command()
@click.option("-f", "--filename", default=".", help="manifest file or dir")
@click.option("-n", "--namespace", default="default")
@click.option("--build", "build_path", default=None,
              help="directory with Dockerfile to upload as build context")
@click.option("--wait/--no-wait", default=False)
def apply(filename, namespace, build_path, wait):
    """Server-side apply substratus manifests (with optional -b upload
    build and readiness wait) — the reference's `sub apply [-b]`."""
    kube = _kube()
    objs = find_manifests(filename)
    if not objs:
        _say("[red]no substratus manifests found[/red]")
        sys.exit(1)
    for obj in objs:
        obj.namespace = namespace
        _apply_with_optional_upload(kube, obj, build_path, wait)


def _stream_pod_logs(namespace: str, selector: str):
    """Background `kubectl logs -f` for the workload pods (the reference
    TUI streams pod logs per role, reference internal/tui/pods.go:222+).
    Returns the Popen or None when kubectl is unavailable."""
    import shutil
    import subprocess
    k = shutil.which("kubectl")
    if k is None:
        return None
    return subprocess.Popen(
        [k, "logs", "-f", "-n", namespace, "-l", selector, "--all-containers",
         "--prefix", "--ignore-errors"],
        stderr=subprocess.DEVNULL)


@main.command()
@click.argument("path", default=".")
@click.option("-n", "--namespace", default="default")
@click.option("-i", "--increment", is_flag=True,
              help="create a new auto-versioned name-N object")
@click.option("-r", "--replace", is_flag=True, help="replace existing object")
@click.option("--logs/--no-logs", default=True,
              help="stream workload pod logs while waiting")
def run(path, namespace, increment, replace, logs):
    """Upload PATH (with Dockerfile) and run it as a Model build+train."""
    kube = _kube()
    objs = find_manifests(path, kind_filter="Model") or \
        find_manifests(path)
    if not objs:
        _say("[red]no substratus manifests found[/red]")
        sys.exit(1)
    from ..tui import select_manifest
    obj = select_manifest(objs)
    obj.namespace = namespace
    if increment:
        obj.name = next_version_name(kube, obj.kind, namespace, obj.name)
        _say(f"auto-versioned name: {obj.name}")
    elif replace:
        kube.delete("substratus.ai/v1", obj.kind, namespace, obj.name)
    log_proc = _stream_pod_logs(namespace, f"model={obj.name}") if logs \
        else None
    try:
        _apply_with_optional_upload(kube, obj, path, wait=True)
    finally:
        if log_proc is not None:
            log_proc.terminate()


@main.command()
@click.argument("path", default=".")
@click.option("-n", "--namespace", default="default")
@click.option("--sync-dir", default=".",
              help="local directory mirrored from the notebook")
@click.option("--port", default=8888)
@click.option("--no-sync", is_flag=True)
def notebook(path, namespace, sync_dir, port, no_sync):
    """Dev loop: build a Notebook from the first manifest in PATH, upload
    the dir as its image context, wait, sync files, port-forward 8888."""
    kube = _kube()
    objs = find_manifests(path)
    if not objs:
        _say("[red]no substratus manifests found[/red]")
        sys.exit(1)
    from ..tui import select_manifest
    nb = sclient.notebook_for_object(select_manifest(objs))
    nb.namespace = namespace
    _apply_with_optional_upload(kube, nb, path, wait=True)

    ns, pod = sclient.pod_for_notebook(nb)
    fwd = sclient.sync.port_forward(ns, pod, port, 8888)
    _say(f"[green]notebook ready[/green] → http://localhost:{port} "
         f"(token: default)")
    stop = threading.Event()
    try:
        if no_sync:
            fwd.wait()
        else:
            sclient.sync_files_from_notebook(
                ns, pod, sync_dir, stop=stop,
                on_event=lambda e: _say(f"  sync {e['op']} {e['path']}"))
    except KeyboardInterrupt:
        pass
    finally:
        stop.set()
        fwd.terminate()


@main.command()
@click.argument("path", default=".")
@click.option("-n", "--namespace", default="default")
def serve(path, namespace):
    """Apply a Server manifest and wait for it to serve."""
    kube = _kube()
    objs = find_manifests(path, kind_filter="Server")
    if not objs:
        _say("[red]no Server manifest found[/red]")
        sys.exit(1)
    for obj in objs:
        obj.namespace = namespace
        _apply_with_optional_upload(kube, obj, None, wait=True)
        _say(f"Server {obj.name} ready on service {obj.name}-server:8080")


@main.command()
@click.argument("kind")
@click.argument("name", required=False)
@click.option("-n", "--namespace", default="default")
@click.option("-o", "--output", default="table",
              type=click.Choice(["table", "json"]))
def get(kind, name, namespace, output):
    """List substratus objects (`sub get models|datasets|servers|
    notebooks [name]`), table or -o yaml."""
    kube = _kube()
    kind = {k.lower(): k for k in KINDS}.get(kind.rstrip("s").lower(), kind)
    if name:
        raw = kube.get("substratus.ai/v1", kind, namespace, name)
        if raw is None:
            _say(f"[red]{kind}/{name} not found[/red]")
            sys.exit(1)
        import json
        print(json.dumps(raw, indent=2))
        return
    rows = kube.list("substratus.ai/v1", kind, namespace)
    if output == "json":
        import json
        print(json.dumps(rows, indent=2))
        return
    for o in rows:
        status = o.get("status") or {}
        ready = status.get("ready", False)
        conds = status.get("conditions") or []
        last = conds[-1] if conds else {}
        cond = f"{last.get('type', '-')}:{last.get('reason', '-')}" \
            if last else "-"
        print(f"{o['metadata']['name']}\tready={ready}\t{cond}")


@main.command()
@click.argument("kind")
@click.argument("name")
@click.option("-n", "--namespace", default="default")
@click.option("-f", "--follow", is_flag=True)
def logs(kind, name, namespace, follow):
    """Stream the workload pod logs of a substratus object."""
    import shutil
    import subprocess
    k = shutil.which("kubectl")
    if k is None:
        _say("[red]kubectl not found[/red]")
        sys.exit(1)
    kind = {kk.lower(): kk for kk in KINDS}.get(kind.rstrip("s").lower(),
                                                kind).lower()
    args = [k, "logs", "-n", namespace, "-l", f"{kind}={name}",
            "--all-containers", "--prefix", "--ignore-errors"]
    if follow:
        args.append("-f")
    subprocess.run(args)


@main.command()
@click.option("-m", "--model", "server_name", required=True,
              help="Server object name")
@click.option("-p", "--prompt", default="Hello")
@click.option("-n", "--namespace", default="default")
@click.option("--max-tokens", default=32)
@click.option("--temperature", default=0.0)
@click.option("--port", default=18080)
def infer(server_name, prompt, namespace, max_tokens, temperature, port):
    """Send a completion request to a Server through a port-forward
    (the reference's `sub infer` exists but is disabled,
    reference internal/cli/root.go:19 — here it works)."""
    import json
    import time
    import urllib.request

    from .. import client as sclient
    fwd = sclient.sync.port_forward(namespace, f"{server_name}-server",
                                    port, 8080, resource="service")
    try:
        time.sleep(2)
        body = json.dumps({"prompt": prompt, "max_tokens": max_tokens,
                           "temperature": temperature}).encode()
        req = urllib.request.Request(
            f"http://localhost:{port}/v1/completions", data=body,
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=300) as r:
            out = json.loads(r.read())
        print(out["choices"][0]["text"])
    finally:
        fwd.terminate()


@main.command()
@click.argument("name")
@click.option("-n", "--namespace", default="default")
def suspend(name, namespace):
    """Suspend a Notebook (deletes its pod, keeps artifacts) — the
    reference TUI's suspend command (reference tui/common.go:271-301)."""
    kube = _kube()
    try:
        kube.patch("substratus.ai/v1", "Notebook", namespace, name,
                   {"spec": {"suspend": True}})
        _say(f"notebook {name} suspended")
    except Exception:
        _say(f"[red]Notebook/{name} not found[/red]")
        sys.exit(1)


@main.command()
@click.argument("name")
@click.option("-n", "--namespace", default="default")
def resume(name, namespace):
    """Resume a suspended Notebook."""
    kube = _kube()
    try:
        kube.patch("substratus.ai/v1", "Notebook", namespace, name,
                   {"spec": {"suspend": False}})
        _say(f"notebook {name} resuming")
    except Exception:
        _say(f"[red]Notebook/{name} not found[/red]")
        sys.exit(1)


@main.command()
@click.argument("kind")
@click.argument("name")
@click.option("-n", "--namespace", default="default")
def delete(kind, name, namespace):
    """Delete a substratus object (`sub delete kind/name`)."""
    kube = _kube()
    kind = {k.lower(): k for k in KINDS}.get(kind.rstrip("s").lower(), kind)
    if kube.delete("substratus.ai/v1", kind, namespace, name):
        _say(f"deleted {kind}/{name}")
    else:
        _say(f"[red]{kind}/{name} not found[/red]")
        sys.exit(1)


def kubectl_notebook():
    """kubectl plugin shim (reference `kubectl notebook` UX): an
    executable named kubectl-notebook on PATH is discovered by kubectl
    as a plugin; it forwards to `sub notebook`."""
    main(["notebook", *sys.argv[1:]], prog_name="kubectl notebook")


def kubectl_applybuild():
    """kubectl plugin shim (reference `kubectl applybuild`): forwards
    to `sub apply` with the build-context upload. Usage:
    kubectl applybuild -f manifest.yaml --build ./dir"""
    main(["apply", *sys.argv[1:]], prog_name="kubectl applybuild")


if __name__ == "__main__":
    main()
