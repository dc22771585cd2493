"""Parallelism: DP (bucketed RCCL all-reduce) + TP (column/row sharding)."""
from . import comm  # noqa: F401
from .ddp import DataParallel  # noqa: F401
from .tp import (  # noqa: F401
    ColumnParallelLinear,
    RowParallelLinear,
    copy_to_tp,
    gather_from_tp,
    reduce_from_tp,
)
