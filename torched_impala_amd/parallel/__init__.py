from torched_impala_amd.parallel.ddp import GradReducer, init_distributed

__all__ = ["GradReducer", "init_distributed"]
