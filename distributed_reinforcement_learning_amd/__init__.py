"""distributed_reinforcement_learning_amd — MI355X-native distributed deep-RL framework.

A from-scratch rebuild of the capabilities of
``chagmgang/distributed_reinforcement_learning`` (TF1 + gRPC, reference at
/root/reference) designed MI355X-first:

* PyTorch-ROCm for autograd orchestration; hand-written HIP/CDNA4 kernels
  (gfx950, MFMA/LDS-tiled) for the hot path; RCCL over xGMI for learner
  data-parallelism.
* CPU actor processes stream unrolled trajectories into lock-free
  shared-memory rings (``parallel.queue``); the learner publishes weights
  through a seqlock shared-memory buffer (``parallel.weights``) instead of
  per-actor RPC pulls (reference: utils.py:6-22).
* Four algorithms, same entry points as the reference:
  A3C (train_a3c.py), IMPALA/V-trace (train_impala.py),
  Ape-X DQN (train_apex.py), R2D2 (train_r2d2.py).
"""

__version__ = "0.1.0"

from distributed_reinforcement_learning_amd.config import load_config, check_properties

__all__ = ["load_config", "check_properties", "__version__"]
