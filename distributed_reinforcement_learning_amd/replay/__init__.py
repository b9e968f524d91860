from distributed_reinforcement_learning_amd.replay.sum_tree import SumTree
from distributed_reinforcement_learning_amd.replay.memory import Memory
from distributed_reinforcement_learning_amd.replay.local_buffer import LocalBuffer

__all__ = ["SumTree", "Memory", "LocalBuffer"]
