from distributed_reinforcement_learning_amd.trainers import a3c, impala, apex, r2d2

__all__ = ["a3c", "impala", "apex", "r2d2"]
