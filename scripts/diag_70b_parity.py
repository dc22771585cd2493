#!/usr/bin/env python3
"""Diagnose the 70b-trunc2 engine-vs-full-forward token divergence:
same run with RB_DECODE_MFMA on and off (subprocess per mode), printing
per-step engine token, reference top-2 tokens and the top-2 logit gap.
GPU box: python scripts/diag_70b_parity.py [mode]"""
import dataclasses
import json
import os
import subprocess
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def run_mode():
    import torch
    from runbooks_amd.models import build_model, get_config
    from runbooks_amd.models.config import register
    from runbooks_amd.serve import Engine

    name = "llama2-70b"
    cfg = dataclasses.replace(get_config(name), num_layers=2,
                              name=f"{name}-trunc2")
    register(cfg)
    m = build_model(cfg.name, dtype=torch.bfloat16, device="cuda:0", seed=7)
    eng = Engine(m, device="cuda:0", kv_blocks=512, seed=3)
    prompt = [11, 99, 5, 42, 7]
    out = eng.generate(list(prompt), max_new_tokens=8)

    ids = list(prompt)
    rows = []
    for step in range(8):
        with torch.no_grad():
            logits = m(torch.tensor([ids], device="cuda:0"))[0, -1].float()
        top2 = torch.topk(logits, 2)
        ref_tok = int(top2.indices[0])
        gap = float(top2.values[0] - top2.values[1])
        eng_tok = out[step] if step < len(out) else -1
        eng_logit = float(logits[eng_tok]) if eng_tok >= 0 else None
        rows.append({"step": step, "engine": eng_tok, "ref": ref_tok,
                     "gap": round(gap, 5),
                     "eng_vs_top": round(float(top2.values[0]) -
                                         (eng_logit or 0.0), 5)})
        ids.append(eng_tok)  # follow the ENGINE's trajectory
    print(json.dumps({"mfma": os.environ.get("RB_DECODE_MFMA", "1"),
                      "rows": rows}), flush=True)


def main():
    if len(sys.argv) > 1 and sys.argv[1] == "child":
        run_mode()
        return
    for mode in ("1", "0"):
        env = dict(os.environ, RB_DECODE_MFMA=mode)
        r = subprocess.run([sys.executable, __file__, "child"], env=env,
                           capture_output=True, text=True, timeout=600)
        print(r.stdout, end="", flush=True)
        if r.returncode != 0:
            print(r.stderr[-800:], flush=True)


if __name__ == "__main__":
    main()
