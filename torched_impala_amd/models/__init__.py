from torched_impala_amd.models.nets import (
    MLPPolicy, ShallowNet, DeepResNet, build_model,
)

__all__ = ["MLPPolicy", "ShallowNet", "DeepResNet", "build_model"]
