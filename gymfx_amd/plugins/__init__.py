"""Plugin registry: six families, contract-compatible with the reference
entry-point groups (/root/reference/setup.py:11-35 and
app/plugin_loader.py:12-48).

Resolution order: built-in registry first (no installation required), then
importlib.metadata entry points (so third-party plugins keep working).
"""
from __future__ import annotations

from typing import Any, Dict, List, Tuple, Type

GROUPS = (
    "data_feed.plugins",
    "broker.plugins",
    "strategy.plugins",
    "preprocessor.plugins",
    "reward.plugins",
    "metrics.plugins",
)

_BUILTIN: Dict[str, Dict[str, str]] = {
    "data_feed.plugins": {
        "default_data_feed": "gymfx_amd.plugins.data_feeds:DefaultDataFeed",
        "synthetic_data_feed": "gymfx_amd.plugins.data_feeds:SyntheticDataFeed",
    },
    "broker.plugins": {
        "default_broker": "gymfx_amd.plugins.brokers:DefaultBroker",
        "oanda_broker": "gymfx_amd.plugins.brokers:OandaBroker",
    },
    "strategy.plugins": {
        "default_strategy": "gymfx_amd.plugins.strategies:DefaultStrategy",
        "direct_fixed_sltp": "gymfx_amd.plugins.strategies:DirectFixedSLTP",
        "direct_atr_sltp": "gymfx_amd.plugins.strategies:DirectAtrSLTP",
    },
    "preprocessor.plugins": {
        "default_preprocessor": "gymfx_amd.plugins.preprocessors:DefaultPreprocessor",
        "feature_window_preprocessor": "gymfx_amd.plugins.preprocessors:FeatureWindowPreprocessor",
    },
    "reward.plugins": {
        "pnl_reward": "gymfx_amd.plugins.rewards:PnlReward",
        "sharpe_reward": "gymfx_amd.plugins.rewards:SharpeReward",
        "dd_penalized_reward": "gymfx_amd.plugins.rewards:DdPenalizedReward",
    },
    "metrics.plugins": {
        "default_metrics": "gymfx_amd.plugins.metrics:DefaultMetrics",
        "trading_metrics": "gymfx_amd.plugins.metrics:TradingMetrics",
    },
}


def _import_target(target: str) -> Type:
    mod_name, _, attr = target.partition(":")
    import importlib

    mod = importlib.import_module(mod_name)
    return getattr(mod, attr)


def load_plugin(plugin_group: str, plugin_name: str) -> Tuple[Type, List[str]]:
    """Load a plugin class + its required parameter keys.

    Same return contract as the reference loader (app/plugin_loader.py:12-48).
    """
    builtin = _BUILTIN.get(plugin_group, {})
    if plugin_name in builtin:
        klass = _import_target(builtin[plugin_name])
        return klass, list(getattr(klass, "plugin_params", {}).keys())
    try:
        from importlib.metadata import entry_points

        group_entries = entry_points().select(group=plugin_group)
        entry = next(ep for ep in group_entries if ep.name == plugin_name)
        klass = entry.load()
        return klass, list(getattr(klass, "plugin_params", {}).keys())
    except StopIteration:
        raise ImportError(
            f"Plugin {plugin_name} not found in group {plugin_group}."
        ) from None


def get_plugin_params(plugin_group: str, plugin_name: str) -> Dict[str, Any]:
    klass, _ = load_plugin(plugin_group, plugin_name)
    return dict(getattr(klass, "plugin_params", {}))


def available_plugins(plugin_group: str) -> List[str]:
    names = set(_BUILTIN.get(plugin_group, {}))
    try:
        from importlib.metadata import entry_points

        names |= {ep.name for ep in entry_points().select(group=plugin_group)}
    except Exception:
        pass
    return sorted(names)
