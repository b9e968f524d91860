from distributed_reinforcement_learning_amd.runtime.graphed import GraphedImpalaStep
from distributed_reinforcement_learning_amd.runtime.replay_graphed import (
    GraphedReplayStep,
)

__all__ = ["GraphedImpalaStep", "GraphedReplayStep"]
