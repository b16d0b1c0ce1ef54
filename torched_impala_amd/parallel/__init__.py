from torched_impala_amd.parallel.ddp import (
    FlatGradReducer, GradReducer, init_distributed,
)

__all__ = ["FlatGradReducer", "GradReducer", "init_distributed"]
