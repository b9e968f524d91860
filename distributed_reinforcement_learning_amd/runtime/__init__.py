from distributed_reinforcement_learning_amd.runtime.graphed import GraphedImpalaStep

__all__ = ["GraphedImpalaStep"]
