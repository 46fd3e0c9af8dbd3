from .params import EnvParams
from .vec_env import VecFxEnv, resolve_device
from .gym_env import GymFxEnv, build_base_observation_space

__all__ = [
    "EnvParams",
    "VecFxEnv",
    "GymFxEnv",
    "build_base_observation_space",
    "resolve_device",
]
