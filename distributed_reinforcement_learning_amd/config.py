"""Config loading + validation.

Same two-tier scheme as the reference: CLI flags carry only ``--job_name`` and
``--task`` (reference train_impala.py:13-20); everything else lives in a JSON
file with one block per algorithm (reference config.json:2,43,103,146),
validated like reference utils.py:34-45 (``check_properties``).

The JSON schema is kept drop-in compatible with the reference so existing
configs can be reused unchanged.
"""

from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import Any, Dict, List

_VALID_REWARD_CLIPPING = ("abs_one", "soft_asymmetric", "none")


def check_properties(data: Dict[str, Any]) -> None:
    """Validate one algorithm block (reference utils.py:34-45 semantics).

    * every per-actor list has length num_actors
    * available_action[i] <= model_output
    * reward_clipping is a known mode
    """
    num_actors = data["num_actors"]
    envs = data["env"]
    available = data["available_action"]
    if len(envs) != num_actors:
        raise ValueError(
            f"len(env)={len(envs)} != num_actors={num_actors}")
    if len(available) != num_actors:
        raise ValueError(
            f"len(available_action)={len(available)} != num_actors={num_actors}")
    model_output = data["model_output"]
    for i, a in enumerate(available):
        if a > model_output:
            raise ValueError(
                f"available_action[{i}]={a} > model_output={model_output}")
    rc = data.get("reward_clipping", "abs_one")
    if rc not in _VALID_REWARD_CLIPPING:
        raise ValueError(f"unknown reward_clipping {rc!r}")


@dataclass
class Config:
    """Typed view over one algorithm block of config.json."""

    algorithm: str
    raw: Dict[str, Any] = field(repr=False)

    # topology
    server_ip: str = "127.0.0.1"
    server_port: int = 8000
    num_actors: int = 1
    env: List[str] = field(default_factory=list)
    available_action: List[int] = field(default_factory=list)

    # model / data shapes
    model_input: List[int] = field(default_factory=lambda: [84, 84, 4])
    model_output: int = 18
    trajectory: int = 20           # unroll length (a3c/impala/apex)
    seq_len: int = 15              # r2d2
    burn_in: int = 7               # r2d2
    lstm_size: int = 256           # impala 256 / r2d2 64
    batch_size: int = 32
    queue_size: int = 128

    # optimization
    discount_factor: float = 0.99
    start_learning_rate: float = 6e-4
    end_learning_rate: float = 0.0
    learning_frame: int = 1_000_000_000
    gradient_clip_norm: float = 40.0
    baseline_loss_coef: float = 1.0
    entropy_coef: float = 0.05
    reward_clipping: str = "abs_one"

    def __getitem__(self, key: str) -> Any:
        return self.raw[key]

    def get(self, key: str, default: Any = None) -> Any:
        return self.raw.get(key, default)


_FIELDS = [
    "server_ip", "server_port", "num_actors", "env", "available_action",
    "model_input", "model_output", "trajectory", "seq_len", "burn_in",
    "lstm_size", "batch_size", "queue_size", "discount_factor",
    "start_learning_rate", "end_learning_rate", "learning_frame",
    "gradient_clip_norm", "baseline_loss_coef", "entropy_coef",
    "reward_clipping",
]


def load_config(path: str, algorithm: str, validate: bool = True) -> Config:
    """Load the ``algorithm`` block from the JSON config at ``path``."""
    with open(path, "r") as f:
        blob = json.load(f)
    if algorithm not in blob:
        raise KeyError(f"no block {algorithm!r} in {path} "
                       f"(have {sorted(blob)})")
    data = blob[algorithm]
    if validate:
        check_properties(data)
    cfg = Config(algorithm=algorithm, raw=data)
    for name in _FIELDS:
        if name in data:
            setattr(cfg, name, data[name])
    return cfg


def default_config_path() -> str:
    """config.json at the repo root (next to the train_* entry points)."""
    here = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    return os.path.join(here, "config.json")
