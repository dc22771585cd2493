"""runbooks_amd.ops — MI355X (gfx950) compute ops.

GPU tensors run on hand-written CDNA4 HIP kernels (csrc/); CPU tensors run
plain-PyTorch fp32 references (the numerics oracles). A GPU tensor with no
built extension raises — there is no silent eager fallback on the GPU.
"""
from ._backend import ext, has_hip, use_hip  # noqa: F401
from .activations import cross_entropy, swiglu  # noqa: F401
from .adamw import FusedAdamW  # noqa: F401
from .attention import (causal_attention, flash_prefill, paged_decode,  # noqa: F401
                        paged_decode_ref, paged_decode_with_operand)
from .kvcache import BLOCK_SIZE, alloc_kv_cache, kv_append, kv_append_ref  # noqa: F401
from .linear import Linear, decode_linear_raw, fast_linear  # noqa: F401
from .norm import RMSNorm, rmsnorm, rmsnorm_ref  # noqa: F401
from .rope import rope, rope_ref, rope_tables  # noqa: F401
from .sampling import sample_tokens  # noqa: F401
