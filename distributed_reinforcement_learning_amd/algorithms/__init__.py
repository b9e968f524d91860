from distributed_reinforcement_learning_amd.algorithms import a2c, vtrace, dqn, burn_in

__all__ = ["a2c", "vtrace", "dqn", "burn_in"]
