"""Training: LoRA fine-tune loop, checkpointing, data."""
from .checkpoint import latest_checkpoint, load_checkpoint, save_checkpoint  # noqa: F401
from .data import (  # noqa: F401
    JsonlTextDataset,
    SyntheticTokens,
    TextDataset,
    data_loader,
)
from .lora import LoRALinear, apply_lora, lora_state_dict, merge_lora  # noqa: F401
from .trainer import TrainConfig, Trainer  # noqa: F401
