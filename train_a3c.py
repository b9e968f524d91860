#!/usr/bin/env python3
"""A3C entry point (parity with reference train_a3c.py).

    python train_a3c.py --spawn                       # Atari conv config
    python train_a3c.py --spawn --algorithm_block a3c_cartpole   # plumbing
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from distributed_reinforcement_learning_amd.trainers.a3c import main

if __name__ == "__main__":
    main()
