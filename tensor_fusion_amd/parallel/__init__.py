from .tp import (ColumnParallelLinear, RowParallelLinear, TPLlama,
                 build_tp_model)

__all__ = ["TPLlama", "build_tp_model", "ColumnParallelLinear",
           "RowParallelLinear"]
