from distributed_reinforcement_learning_amd.utils.logging import SummaryWriter
from distributed_reinforcement_learning_amd.utils.trajectory import (
    UnrolledA3CTrajectory,
    UnrolledTrajectory,
)
from distributed_reinforcement_learning_amd.utils.timing import StageTimer

__all__ = [
    "SummaryWriter",
    "UnrolledA3CTrajectory",
    "UnrolledTrajectory",
    "StageTimer",
]
