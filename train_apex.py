#!/usr/bin/env python3
"""Ape-X DQN entry point (parity with reference train_apex.py)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from distributed_reinforcement_learning_amd.trainers.apex import main

if __name__ == "__main__":
    main()
