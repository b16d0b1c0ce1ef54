from torched_impala_amd.transport.ringbuffer import (
    SlotQueue, TrajectoryBuffers,
)
from torched_impala_amd.transport.weights import WeightMirror

__all__ = ["SlotQueue", "TrajectoryBuffers", "WeightMirror"]
