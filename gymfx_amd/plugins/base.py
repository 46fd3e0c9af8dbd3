"""Shared plugin shape: ``plugin_params`` defaults + __init__(config) +
set_params(**kwargs) (the contract every reference plugin follows, e.g.
/root/reference/reward_plugins/pnl_reward.py:12-24)."""
from __future__ import annotations

from typing import Any, Dict, Optional


class PluginBase:
    plugin_params: Dict[str, Any] = {}

    def __init__(self, config: Optional[Dict[str, Any]] = None):
        self.params = dict(self.plugin_params)
        if config:
            self.set_params(**config)

    def set_params(self, **kwargs: Any) -> None:
        self.params.update(kwargs)

    def _resolve(self, config: Dict[str, Any]) -> Dict[str, Any]:
        """Merged view: plugin defaults overridden by env config (env wins)."""
        merged = dict(self.params)
        for k in self.plugin_params:
            if k in config and config[k] is not None:
                merged[k] = config[k]
        return merged
