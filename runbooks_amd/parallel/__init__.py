"""Parallelism: DP (bucketed RCCL all-reduce) + TP (column/row sharding)."""
from . import comm  # noqa: F401
from .ddp import DataParallel  # noqa: F401
from .tp import ColumnParallelLinear, RowParallelLinear, copy_to_tp, reduce_from_tp  # noqa: F401
