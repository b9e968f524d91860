from distributed_reinforcement_learning_amd.envs.wrappers import (
    make_uint8_env,
    make_float_env,
    make_uint8_env_no_fire,
    pomdp_uint8_env,
    make_env,
)
from distributed_reinforcement_learning_amd.envs.cartpole import CartPoleEnv
from distributed_reinforcement_learning_amd.envs.synthetic import SyntheticAtariEnv

__all__ = [
    "make_uint8_env", "make_float_env", "make_uint8_env_no_fire",
    "pomdp_uint8_env", "make_env", "CartPoleEnv", "SyntheticAtariEnv",
]
