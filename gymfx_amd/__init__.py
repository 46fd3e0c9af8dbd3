"""gymfx_amd — MI355X-native vectorized FX-trading RL framework.

Capabilities of harveybc/gym-fx (env API + six plugin contracts), re-designed
MI355X-first: SoA vectorized envs in HBM, fused HIP step kernels, MFMA PPO,
RCCL data parallelism.  See SURVEY.md for the reference layer map.
"""
from __future__ import annotations

from typing import Any, Dict

__version__ = "0.1.0"


def build_environment(
    *,
    config: Dict[str, Any],
    data_feed_plugin,
    broker_plugin,
    strategy_plugin,
    preprocessor_plugin,
    reward_plugin,
    metrics_plugin,
):
    """Engine dispatch (parity: /root/reference/gym_fx/__init__.py:4-12).

    ``simulation_engine`` selects the engine: "vectorized" (default, native)
    builds the single-env Gymnasium wrapper over the vectorized engine.  The
    reference's "backtrader"/"nautilus" names are accepted as aliases for the
    native engine (their semantics are what the native engine implements).
    """
    engine = str(config.get("simulation_engine", "vectorized")).lower()
    if engine not in {"vectorized", "backtrader", "nautilus"}:
        raise ValueError(f"unknown simulation_engine '{engine}'")
    from .envs.gym_env import GymFxEnv

    return GymFxEnv(
        config=config,
        data_feed_plugin=data_feed_plugin,
        broker_plugin=broker_plugin,
        strategy_plugin=strategy_plugin,
        preprocessor_plugin=preprocessor_plugin,
        reward_plugin=reward_plugin,
        metrics_plugin=metrics_plugin,
    )


def build_vec_environment(config: Dict[str, Any], market_data=None, **kwargs):
    """Build the N-env vectorized engine directly (training path)."""
    from .envs.vec_env import VecFxEnv
    from .plugins import load_plugin

    if market_data is None:
        feed_cls, _ = load_plugin(
            "data_feed.plugins", str(config.get("data_feed_plugin", "default_data_feed"))
        )
        market_data = feed_cls(config).load_data(config)
    return VecFxEnv(config, market_data, **kwargs)
