"""Load HuggingFace-format checkpoints into the native Transformer.

The import path of the platform: the model-loader image downloads HF
weights into /content/artifacts (reference
examples/facebook-opt-125m/base-model.yaml:7-9, SURVEY.md §2b
"model-loader image"), and the trainer/server load them from
/content/model. This module maps the llama / falcon / OPT HF naming onto
the native module tree, splits falcon's fused query_key_value, and
TP-shards each tensor for the local rank.
"""
from __future__ import annotations

import json
from pathlib import Path

import torch

from .config import ModelConfig
from .transformer import Transformer


def detect_family(cfg: ModelConfig) -> str:
    if cfg.pos == "learned":
        return "gpt2" if cfg.act == "gelu_tanh" else "opt"
    if cfg.parallel_residual:
        return "falcon"
    return "llama"


# ---------------------------------------------------------------------------
# HF name -> native name
# ---------------------------------------------------------------------------

def _map_llama(name: str):
    n = name
    n = n.replace("model.embed_tokens", "embed")
    n = n.replace("model.layers.", "blocks.")
    n = n.replace(".self_attn.", ".attn.")
    n = n.replace(".input_layernorm.", ".norm1.")
    n = n.replace(".post_attention_layernorm.", ".norm2.")
    n = n.replace("model.norm", "norm_f")
    return n


def _map_opt(name: str):
    n = name
    n = n.replace("model.decoder.embed_tokens", "embed")
    n = n.replace("model.decoder.embed_positions", "embed_pos")
    n = n.replace("model.decoder.layers.", "blocks.")
    n = n.replace(".self_attn.", ".attn.")
    n = n.replace(".out_proj.", ".o_proj.")
    n = n.replace(".self_attn_layer_norm.", ".norm1.")
    n = n.replace("model.decoder.final_layer_norm", "norm_f")
    n = n.replace(".final_layer_norm.", ".norm2.")
    n = n.replace(".fc1.", ".mlp.fc1.")
    n = n.replace(".fc2.", ".mlp.down_proj.")
    return n


def _map_gpt2(name: str):
    n = name
    if not n.startswith(("transformer.", "lm_head")):
        n = "transformer." + n          # GPT2Model vs GPT2LMHeadModel
    n = n.replace("transformer.wte", "embed")
    n = n.replace("transformer.wpe", "embed_pos")
    n = n.replace("transformer.h.", "blocks.")
    n = n.replace(".attn.c_proj.", ".attn.o_proj.")
    n = n.replace(".mlp.c_fc.", ".mlp.fc1.")
    n = n.replace(".mlp.c_proj.", ".mlp.down_proj.")
    n = n.replace(".ln_1.", ".norm1.")
    n = n.replace(".ln_2.", ".norm2.")
    n = n.replace("transformer.ln_f", "norm_f")
    return n


def _map_falcon(name: str):
    n = name
    n = n.replace("transformer.word_embeddings", "embed")
    n = n.replace("transformer.h.", "blocks.")
    n = n.replace(".self_attention.dense.", ".attn.o_proj.")
    n = n.replace(".mlp.dense_h_to_4h.", ".mlp.fc1.")
    n = n.replace(".mlp.dense_4h_to_h.", ".mlp.down_proj.")
    # falcon-7b single-norm / falcon-40b two-norm naming
    n = n.replace(".input_layernorm.", ".norm1.")
    n = n.replace(".ln_attn.", ".norm1.")
    n = n.replace(".ln_mlp.", ".norm2.")
    n = n.replace("transformer.ln_f", "norm_f")
    return n


def _split_falcon_qkv(w: torch.Tensor, cfg: ModelConfig):
    """Falcon fuses QKV as [n_kv_groups, (g_q + 2), head_dim, hidden] with
    the group's q heads followed by its k and v head."""
    g = cfg.num_heads // cfg.num_kv_heads
    dh = cfg.head_dim
    w = w.view(cfg.num_kv_heads, g + 2, dh, cfg.hidden_size)
    q = w[:, :g].reshape(cfg.num_heads * dh, cfg.hidden_size)
    k = w[:, g].reshape(cfg.num_kv_heads * dh, cfg.hidden_size)
    v = w[:, g + 1].reshape(cfg.num_kv_heads * dh, cfg.hidden_size)
    return q, k, v


def convert_hf_state_dict(hf_state: dict, cfg: ModelConfig) -> dict:
    family = detect_family(cfg)
    out: dict[str, torch.Tensor] = {}
    mapper = {"llama": _map_llama, "opt": _map_opt,
              "falcon": _map_falcon, "gpt2": _map_gpt2}[family]
    for name, w in hf_state.items():
        if name.endswith(".rotary_emb.inv_freq"):
            continue
        if family == "gpt2":
            if name.endswith((".attn.bias", ".attn.masked_bias")):
                continue  # causal-mask buffers, not weights
            if ".attn.c_attn." in name:
                # Conv1D fused QKV: weight [in, 3*out] -> three [out, in]
                blk = name.split(".")[name.split(".").index("h") + 1]
                if name.endswith(".weight"):
                    qw, kw, vw = w.t().chunk(3, dim=0)
                else:
                    qw, kw, vw = w.chunk(3, dim=0)
                kind = "weight" if name.endswith(".weight") else "bias"
                out[f"blocks.{blk}.attn.q_proj.{kind}"] = qw.contiguous()
                out[f"blocks.{blk}.attn.k_proj.{kind}"] = kw.contiguous()
                out[f"blocks.{blk}.attn.v_proj.{kind}"] = vw.contiguous()
                continue
            if name.endswith(".weight") and w.dim() == 2 and any(
                    k in name for k in (".c_proj.", ".c_fc.")):
                w = w.t().contiguous()  # Conv1D stores [in, out]
        if family == "falcon" and ".self_attention.query_key_value." in name:
            blk = name.split(".")[2 if name.startswith("transformer") else 1]
            q, k, v = _split_falcon_qkv(w, cfg)
            out[f"blocks.{blk}.attn.q_proj.weight"] = q
            out[f"blocks.{blk}.attn.k_proj.weight"] = k
            out[f"blocks.{blk}.attn.v_proj.weight"] = v
            continue
        n = mapper(name)
        if family == "opt" and n.startswith("embed_pos"):
            # OPT's learned positions carry a +2 offset (HF
            # OPTLearnedPositionalEmbedding); drop the two pad rows.
            w = w[2:]
        out[n] = w
    # gemma stores RMSNorm weights zero-centered (x * (1 + w)); fold the
    # +1 at import so the fused rmsnorm kernel stays family-agnostic
    if cfg.act == "gelu_glu" and cfg.norm == "rmsnorm":
        for k in list(out):
            if k.endswith((".norm1.weight", ".norm2.weight")) or                     k == "norm_f.weight":
                out[k] = out[k].float() + 1.0
    # tied embeddings: derive lm_head when absent
    if "lm_head.weight" not in out and "embed.weight" in out:
        out["lm_head.weight"] = out["embed.weight"]
    return out


# ---------------------------------------------------------------------------
# TP sharding
# ---------------------------------------------------------------------------

def shard_state_dict(state: dict, model: Transformer, rank: int,
                     tp: int) -> dict:
    """Slice full tensors to this rank's shard, matching parallel/tp.py's
    column (output-dim) / row (input-dim) split."""
    if tp <= 1:
        return state
    out = {}
    named = dict(model.named_parameters())
    named.update(dict(model.named_buffers()))
    for k, w in state.items():
        tgt = named.get(k)
        if tgt is None or tgt.shape == w.shape:
            out[k] = w
            continue
        if tgt.shape[0] != w.shape[0]:        # column-parallel (incl. bias)
            s = tgt.shape[0]
            out[k] = w[rank * s:(rank + 1) * s]
        elif w.dim() == 2 and tgt.shape[1] != w.shape[1]:  # row-parallel
            s = tgt.shape[1]
            out[k] = w[:, rank * s:(rank + 1) * s]
        else:
            out[k] = w
    return out


# ---------------------------------------------------------------------------
# Entry points
# ---------------------------------------------------------------------------

def load_safetensors_dir(path: str | Path) -> dict:
    from safetensors.torch import load_file
    path = Path(path)
    files = sorted(path.glob("*.safetensors"))
    if not files:
        raise FileNotFoundError(f"no .safetensors under {path}")
    state: dict[str, torch.Tensor] = {}
    for f in files:
        state.update(load_file(str(f)))
    return state


def config_from_hf_json(path: str | Path) -> ModelConfig:
    """Derive a ModelConfig from an HF config.json (for models not in the
    registry)."""
    d = json.loads(Path(path).read_text())
    mt = d.get("model_type", "llama")
    if mt == "gemma":
        heads = d["num_attention_heads"]
        return ModelConfig(
            d.get("_name_or_path", "hf-gemma"),
            vocab_size=d["vocab_size"], hidden_size=d["hidden_size"],
            num_layers=d["num_hidden_layers"], num_heads=heads,
            num_kv_heads=d.get("num_key_value_heads", heads),
            intermediate_size=d["intermediate_size"],
            max_seq_len=d.get("max_position_embeddings", 8192),
            head_dim=d.get("head_dim", d["hidden_size"] // heads),
            norm_eps=d.get("rms_norm_eps", 1e-6),
            rope_theta=d.get("rope_theta", 10000.0),
            act="gelu_glu", tie_embeddings=True,
            embed_scale=d["hidden_size"] ** 0.5)
    if mt == "qwen2":
        return ModelConfig(
            d.get("_name_or_path", "hf-qwen2"),
            vocab_size=d["vocab_size"], hidden_size=d["hidden_size"],
            num_layers=d["num_hidden_layers"],
            num_heads=d["num_attention_heads"],
            num_kv_heads=d.get("num_key_value_heads",
                               d["num_attention_heads"]),
            intermediate_size=d["intermediate_size"],
            max_seq_len=d.get("max_position_embeddings", 32768),
            norm_eps=d.get("rms_norm_eps", 1e-6),
            rope_theta=d.get("rope_theta", 1000000.0),
            qkv_bias=True,
            tie_embeddings=d.get("tie_word_embeddings", False))
    if mt in ("llama", "mistral"):
        return ModelConfig(
            d.get("_name_or_path", "hf-llama"),
            vocab_size=d["vocab_size"], hidden_size=d["hidden_size"],
            num_layers=d["num_hidden_layers"],
            num_heads=d["num_attention_heads"],
            num_kv_heads=d.get("num_key_value_heads",
                               d["num_attention_heads"]),
            intermediate_size=d["intermediate_size"],
            max_seq_len=d.get("max_position_embeddings", 4096),
            norm_eps=d.get("rms_norm_eps", 1e-5),
            rope_theta=d.get("rope_theta", 10000.0))
    if mt in ("falcon", "RefinedWeb", "RefinedWebModel"):
        heads = d["num_attention_heads"]
        return ModelConfig(
            d.get("_name_or_path", "hf-falcon"),
            vocab_size=d["vocab_size"], hidden_size=d["hidden_size"],
            num_layers=d["num_hidden_layers"], num_heads=heads,
            num_kv_heads=d.get("num_kv_heads",
                               heads if not d.get("multi_query", True) else 1),
            intermediate_size=4 * d["hidden_size"],
            head_dim=d["hidden_size"] // heads,
            norm="layernorm", act="gelu", parallel_residual=True,
            tie_embeddings=True)
    if mt == "gpt2":
        return ModelConfig(
            d.get("_name_or_path", "hf-gpt2"),
            vocab_size=d["vocab_size"], hidden_size=d["n_embd"],
            num_layers=d["n_layer"], num_heads=d["n_head"],
            num_kv_heads=d["n_head"],
            intermediate_size=d.get("n_inner") or 4 * d["n_embd"],
            max_seq_len=d.get("n_positions", 1024),
            norm="layernorm", act="gelu_tanh", pos="learned",
            tie_embeddings=True, mlp_bias=True, attn_bias=True)
    if mt == "opt":
        return ModelConfig(
            d.get("_name_or_path", "hf-opt"),
            vocab_size=d["vocab_size"], hidden_size=d["hidden_size"],
            num_layers=d["num_hidden_layers"],
            num_heads=d["num_attention_heads"],
            num_kv_heads=d["num_attention_heads"],
            intermediate_size=d["ffn_dim"],
            max_seq_len=d.get("max_position_embeddings", 2048),
            norm="layernorm", act="relu", pos="learned",
            tie_embeddings=True, mlp_bias=True, attn_bias=True)
    raise ValueError(f"unsupported model_type {mt}")


def load_pretrained(model: Transformer, path: str | Path, rank: int = 0,
                    tp: int = 1, strict: bool = True) -> Transformer:
    """Load an HF-format checkpoint directory into `model` (TP-sharded)."""
    hf = load_safetensors_dir(path)
    state = convert_hf_state_dict(hf, model.cfg)
    state = shard_state_dict(state, model, rank, tp)
    state = {k: v.to(model.dtype) if v.is_floating_point() else v
             for k, v in state.items()}
    missing, unexpected = model.load_state_dict(state, strict=False)
    unexpected = [u for u in unexpected if "rotary" not in u]
    missing = [m for m in missing if not m.startswith("rope_")]
    if strict and (missing or unexpected):
        raise KeyError(f"load_pretrained: missing={missing[:5]} "
                       f"unexpected={unexpected[:5]}")
    return model
