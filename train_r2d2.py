#!/usr/bin/env python3
"""R2D2 entry point (parity with reference train_r2d2.py)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from distributed_reinforcement_learning_amd.trainers.r2d2 import main

if __name__ == "__main__":
    main()
