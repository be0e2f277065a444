from .tp import ParallelContext, ColumnParallelLinear, RowParallelLinear

__all__ = ["ParallelContext", "ColumnParallelLinear", "RowParallelLinear"]
