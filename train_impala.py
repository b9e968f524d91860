#!/usr/bin/env python3
"""IMPALA entry point (parity with reference train_impala.py).

Launch one process per role, e.g.:
    python train_impala.py --job_name learner
    python train_impala.py --job_name actor --task 0
or everything at once:
    python train_impala.py --spawn
Multi-GPU learner (one rank per MI355X, RCCL all-reduce):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 train_impala.py --job_name learner --spawn
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from distributed_reinforcement_learning_amd.trainers.impala import main

if __name__ == "__main__":
    main()
