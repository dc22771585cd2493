"""Model configurations for the families the reference platform serves.

The reference orchestrates external HuggingFace images for these models
(reference examples/: facebook-opt-125m, llama2-7b, llama2-70b,
falcon-7b-instruct, falcon-40b — SURVEY.md §2b). Here they are first-class
configs of our own runtime.
"""
from __future__ import annotations

from dataclasses import dataclass


@dataclass
class ModelConfig:
    name: str
    vocab_size: int
    hidden_size: int
    num_layers: int
    num_heads: int
    num_kv_heads: int
    intermediate_size: int
    max_seq_len: int = 4096
    head_dim: int | None = None
    norm: str = "rmsnorm"            # rmsnorm | layernorm
    norm_eps: float = 1e-5
    act: str = "silu_glu"            # silu_glu | gelu | relu
    pos: str = "rope"                # rope | learned
    rope_theta: float = 10000.0
    parallel_residual: bool = False  # falcon-style attn+mlp parallel block
    single_norm: bool = False        # falcon-7b: one shared ln for attn+mlp
    # sliding-window attention width (mistral). Enforced by the serving
    # engine at BLOCK_SIZE granularity (up to BLOCK_SIZE-1 extra tokens
    # of context vs the strict window); expired KV blocks are freed.
    sliding_window: int | None = None
    tie_embeddings: bool = False
    mlp_bias: bool = False
    attn_bias: bool = False   # bias on q/k/v AND o_proj (OPT)
    qkv_bias: bool = False    # bias on q/k/v only (qwen2)
    embed_scale: float = 1.0  # gemma multiplies embeddings by sqrt(hidden)

    def __post_init__(self):
        if self.head_dim is None:
            self.head_dim = self.hidden_size // self.num_heads

    @property
    def params_b(self) -> float:
        """Rough parameter count in billions."""
        h, l, v = self.hidden_size, self.num_layers, self.vocab_size
        kvh = self.num_kv_heads * self.head_dim
        attn = h * (self.num_heads * self.head_dim) + 2 * h * kvh + \
            self.num_heads * self.head_dim * h
        mlp = (3 if self.act == "silu_glu" else 2) * h * self.intermediate_size
        emb = v * h * (1 if self.tie_embeddings else 2)
        return (l * (attn + mlp) + emb) / 1e9


_REGISTRY: dict[str, ModelConfig] = {}


def register(cfg: ModelConfig) -> ModelConfig:
    _REGISTRY[cfg.name] = cfg
    return cfg


def get_config(name: str) -> ModelConfig:
    if name not in _REGISTRY:
        raise KeyError(f"unknown model config '{name}'; have {sorted(_REGISTRY)}")
    return _REGISTRY[name]


def list_configs() -> list[str]:
    return sorted(_REGISTRY)


# --- llama2 family (RMSNorm, RoPE, SwiGLU) ---------------------------------
register(ModelConfig("llama2-7b", vocab_size=32000, hidden_size=4096,
                     num_layers=32, num_heads=32, num_kv_heads=32,
                     intermediate_size=11008))
register(ModelConfig("llama2-13b", vocab_size=32000, hidden_size=5120,
                     num_layers=40, num_heads=40, num_kv_heads=40,
                     intermediate_size=13824))
register(ModelConfig("llama2-70b", vocab_size=32000, hidden_size=8192,
                     num_layers=80, num_heads=64, num_kv_heads=8,
                     intermediate_size=28672))

register(ModelConfig("llama3-8b", vocab_size=128256, hidden_size=4096,
                     num_layers=32, num_heads=32, num_kv_heads=8,
                     intermediate_size=14336, max_seq_len=8192,
                     rope_theta=500000.0))
register(ModelConfig("llama3-70b", vocab_size=128256, hidden_size=8192,
                     num_layers=80, num_heads=64, num_kv_heads=8,
                     intermediate_size=28672, max_seq_len=8192,
                     rope_theta=500000.0))
# mistral-7b: llama-shaped GQA-8 with a 4k sliding window (the engine
# enforces it block-aligned and frees expired KV blocks)
register(ModelConfig("mistral-7b", vocab_size=32000, hidden_size=4096,
                     num_layers=32, num_heads=32, num_kv_heads=8,
                     intermediate_size=14336, max_seq_len=8192,
                     rope_theta=1000000.0, sliding_window=4096))

# --- qwen2 family (llama-shaped + q/k/v bias, GQA, huge vocab) -------------
register(ModelConfig("qwen2-7b", vocab_size=152064, hidden_size=3584,
                     num_layers=28, num_heads=28, num_kv_heads=4,
                     intermediate_size=18944, max_seq_len=32768,
                     norm_eps=1e-6, rope_theta=1000000.0, qkv_bias=True))

# --- gemma family (GeGLU, RMSNorm(1+w) folded at load, scaled embed) -------
register(ModelConfig("gemma-7b", vocab_size=256000, hidden_size=3072,
                     num_layers=28, num_heads=16, num_kv_heads=16,
                     intermediate_size=24576, max_seq_len=8192, head_dim=256,
                     norm_eps=1e-6, act="gelu_glu", tie_embeddings=True,
                     embed_scale=3072 ** 0.5))

# --- falcon family (LayerNorm, RoPE, GELU, parallel residual, MQA/GQA) -----
register(ModelConfig("falcon-7b", vocab_size=65024, hidden_size=4544,
                     num_layers=32, num_heads=71, num_kv_heads=1,
                     intermediate_size=4 * 4544, head_dim=64,
                     norm="layernorm", act="gelu", parallel_residual=True,
                     single_norm=True, tie_embeddings=True))
register(ModelConfig("falcon-40b", vocab_size=65024, hidden_size=8192,
                     num_layers=60, num_heads=128, num_kv_heads=8,
                     intermediate_size=4 * 8192, head_dim=64,
                     norm="layernorm", act="gelu", parallel_residual=True,
                     tie_embeddings=True))

# --- GPT-2 family (LayerNorm, learned positions, tanh-GELU, biases) --------
register(ModelConfig("gpt2", vocab_size=50257, hidden_size=768,
                     num_layers=12, num_heads=12, num_kv_heads=12,
                     intermediate_size=3072, max_seq_len=1024,
                     norm="layernorm", act="gelu_tanh", pos="learned",
                     tie_embeddings=True, mlp_bias=True, attn_bias=True))
register(ModelConfig("gpt2-xl", vocab_size=50257, hidden_size=1600,
                     num_layers=48, num_heads=25, num_kv_heads=25,
                     intermediate_size=6400, max_seq_len=1024,
                     norm="layernorm", act="gelu_tanh", pos="learned",
                     tie_embeddings=True, mlp_bias=True, attn_bias=True))

# --- OPT family (LayerNorm, learned positions, ReLU) -----------------------
register(ModelConfig("opt-125m", vocab_size=50272, hidden_size=768,
                     num_layers=12, num_heads=12, num_kv_heads=12,
                     intermediate_size=3072, max_seq_len=2048,
                     norm="layernorm", act="relu", pos="learned",
                     tie_embeddings=True, mlp_bias=True, attn_bias=True))

# --- small GPU smoke config (head_dim 64 => decode-kernel compatible) ------
register(ModelConfig("smoke-llama", vocab_size=512, hidden_size=256,
                     num_layers=2, num_heads=4, num_kv_heads=2,
                     intermediate_size=512, max_seq_len=256))

# --- small GPU smoke config for the gemma family (Dh=256 kernels) ----------
register(ModelConfig("smoke-gemma", vocab_size=512, hidden_size=512,
                     num_layers=2, num_heads=2, num_kv_heads=2,
                     intermediate_size=1024, max_seq_len=256, head_dim=256,
                     norm_eps=1e-6, act="gelu_glu", tie_embeddings=True,
                     embed_scale=512 ** 0.5))

# --- tiny configs for CPU tests --------------------------------------------
register(ModelConfig("tiny-llama", vocab_size=256, hidden_size=64,
                     num_layers=2, num_heads=4, num_kv_heads=2,
                     intermediate_size=128, max_seq_len=128))
register(ModelConfig("tiny-falcon", vocab_size=256, hidden_size=64,
                     num_layers=2, num_heads=8, num_kv_heads=1,
                     intermediate_size=128, head_dim=8, max_seq_len=128,
                     norm="layernorm", act="gelu", parallel_residual=True,
                     single_norm=True, tie_embeddings=True))
# 8 heads / 8 kv-heads so CPU torchrun rehearsals of the driver's TP=8
# scale command work on gloo (tiny-llama's 4 heads can't split 8 ways)
register(ModelConfig("tiny-llama-8h", vocab_size=256, hidden_size=64,
                     num_layers=2, num_heads=8, num_kv_heads=8,
                     intermediate_size=128, head_dim=8, max_seq_len=128))
register(ModelConfig("tiny-qwen", vocab_size=256, hidden_size=64,
                     num_layers=2, num_heads=4, num_kv_heads=2,
                     intermediate_size=128, max_seq_len=128,
                     norm_eps=1e-6, qkv_bias=True))
register(ModelConfig("tiny-gemma", vocab_size=256, hidden_size=64,
                     num_layers=2, num_heads=4, num_kv_heads=2,
                     intermediate_size=128, head_dim=16, max_seq_len=128,
                     norm_eps=1e-6, act="gelu_glu", tie_embeddings=True,
                     embed_scale=8.0))
register(ModelConfig("tiny-gpt2", vocab_size=256, hidden_size=64,
                     num_layers=2, num_heads=4, num_kv_heads=4,
                     intermediate_size=128, max_seq_len=128,
                     norm="layernorm", act="gelu_tanh", pos="learned",
                     tie_embeddings=True, mlp_bias=True, attn_bias=True))
register(ModelConfig("tiny-opt", vocab_size=256, hidden_size=64,
                     num_layers=2, num_heads=4, num_kv_heads=4,
                     intermediate_size=128, max_seq_len=128,
                     norm="layernorm", act="relu", pos="learned",
                     tie_embeddings=True, mlp_bias=True, attn_bias=True))
