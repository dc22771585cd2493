"""Model families: llama2 / falcon / OPT decoder transformers."""
from .config import ModelConfig, get_config, list_configs, register  # noqa: F401
from .transformer import Attention, Block, Transformer, build_model  # noqa: F401
