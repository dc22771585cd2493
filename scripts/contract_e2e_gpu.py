#!/usr/bin/env python3
"""In-pod container-contract E2E on a real GPU: the exact flow the
controllers orchestrate (reference docs/container-contract.md +
test/system.sh), run against /tmp/content dirs instead of bucket
mounts:

  1. model-loader  -> writes base model to  .../base/artifacts
  2. dataset-loader-> writes data to        .../data/artifacts
  3. trainer       -> PARAM_* env, reads model+data mounts, trains N
                      steps on cuda, checkpoints to .../ft/artifacts
  4. server        -> serves the fine-tuned artifacts on :8080,
                      /v1/completions answers (readiness contract GET /)

GPU box: python scripts/contract_e2e_gpu.py
"""
import json
import os
import shutil
import subprocess
import sys
import threading
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

BASE = "/tmp/contract_e2e"


def run_main(mod: str, env: dict, timeout: int = 300) -> None:
    e = dict(os.environ, **env)
    r = subprocess.run([sys.executable, "-m", f"runbooks_amd.workloads.{mod}"],
                       env=e, cwd=ROOT, timeout=timeout,
                       capture_output=True, text=True)
    if r.returncode != 0:
        print(r.stdout[-1500:], r.stderr[-1500:])
        raise SystemExit(f"{mod} failed rc={r.returncode}")


def main():
    shutil.rmtree(BASE, ignore_errors=True)
    base_art = f"{BASE}/base/artifacts"
    data_art = f"{BASE}/data/artifacts"
    ft_art = f"{BASE}/ft/artifacts"
    for d in (base_art, data_art, ft_art):
        os.makedirs(d)

    # 1. model-loader (random init of a registry model: no network)
    run_main("model_loader", {
        "ARTIFACTS_DIR": base_art, "PARAM_NAME": "smoke-llama",
        "PARAM_SYNTHETIC": "true"})
    assert os.path.exists(f"{base_art}/model.safetensors") or any(
        f.endswith(".safetensors") for f in os.listdir(base_art)), \
        os.listdir(base_art)
    print(json.dumps({"step": "model-loader", "ok": True,
                      "files": sorted(os.listdir(base_art))}), flush=True)

    # 2. dataset-loader (synthetic jsonl)
    run_main("dataset_loader", {
        "ARTIFACTS_DIR": data_art, "PARAM_SYNTHETIC": "true"})
    print(json.dumps({"step": "dataset-loader", "ok": True,
                      "files": sorted(os.listdir(data_art))}), flush=True)

    # 3. trainer: mounts = base model RO + data RO, artifacts RW
    run_main("trainer_main", {
        "MODEL_DIR": base_art, "DATA_DIR": data_art,
        "ARTIFACTS_DIR": ft_art,
        "PARAM_NUM_TRAIN_STEPS": "3", "PARAM_SAVE_STEPS": "3",
        "PARAM_PER_DEVICE_TRAIN_BATCH_SIZE": "2", "PARAM_SEQ_LEN": "32"},
        timeout=600)
    assert os.path.exists(f"{ft_art}/completed.json"), os.listdir(ft_art)
    print(json.dumps({"step": "trainer", "ok": True,
                      "files": sorted(os.listdir(ft_art))[:8]}), flush=True)

    # 4. server on :8080 (background), contract probes
    env = dict(os.environ, MODEL_DIR=ft_art, PORT="18080")
    proc = subprocess.Popen(
        [sys.executable, "-m", "runbooks_amd.workloads.server_main"],
        env=env, cwd=ROOT, stdout=subprocess.DEVNULL,
        stderr=subprocess.DEVNULL)
    try:
        import httpx
        ok = False
        for _ in range(240):
            try:
                if httpx.get("http://127.0.0.1:18080/",
                             timeout=1).status_code == 200:
                    ok = True
                    break
            except Exception:
                time.sleep(0.5)
        assert ok, "server readiness probe (GET /) never succeeded"
        r = httpx.post("http://127.0.0.1:18080/v1/completions", json={
            "model": "model", "prompt": "hello world", "max_tokens": 8,
            "temperature": 0.0}, timeout=120)
        assert r.status_code == 200, r.text[:300]
        body = r.json()
        assert body["usage"]["completion_tokens"] >= 1
        print(json.dumps({"step": "server", "ok": True,
                          "completion_tokens":
                              body["usage"]["completion_tokens"],
                          "text_preview": body["choices"][0]["text"][:40]}),
              flush=True)
    finally:
        proc.terminate()
        proc.wait(timeout=20)
    print(json.dumps({"contract_e2e": "PASS"}), flush=True)


if __name__ == "__main__":
    main()
