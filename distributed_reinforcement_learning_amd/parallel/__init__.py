from distributed_reinforcement_learning_amd.parallel.queue import (
    TrajectoryRing, TrajectoryQueue, queue_schema_for,
)
from distributed_reinforcement_learning_amd.parallel.weights import (
    WeightPublisher, WeightSubscriber,
)
from distributed_reinforcement_learning_amd.parallel.dist import (
    init_distributed, FlatAllReducer, is_distributed, world_size, rank,
)
from distributed_reinforcement_learning_amd.parallel.heartbeat import HeartbeatMonitor

__all__ = [
    "TrajectoryRing", "TrajectoryQueue", "queue_schema_for",
    "WeightPublisher", "WeightSubscriber",
    "init_distributed", "FlatAllReducer", "is_distributed", "world_size",
    "rank", "HeartbeatMonitor",
]
