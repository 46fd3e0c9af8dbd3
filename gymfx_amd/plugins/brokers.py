"""Broker plugins: simulated margin broker parameters + gated live stub.

Parity: /root/reference/broker_plugins/default_broker.py:35-53 (cash, %
commission on notional, % slippage per fill, leverage; long<->short flip =
close+open with two commissions) and oanda_broker.py:42-63 (live gate).
The MI355X engine has no backtrader BackBroker object — the broker IS the
fill arithmetic inside the fused step kernel; this plugin resolves its
parameter pack.
"""
from __future__ import annotations

import os
from typing import Any, Dict

from .base import PluginBase


class DefaultBroker(PluginBase):
    plugin_params = {
        "initial_cash": 10000.0,
        "commission": 0.0,      # fraction of notional per side
        "slippage_perc": 0.0,   # fraction of price applied per fill
        "leverage": 1.0,
    }

    def broker_params(self, config: Dict[str, Any]) -> Dict[str, float]:
        p = self._resolve(config)
        slip = config.get("slippage_perc", config.get("slippage", p["slippage_perc"]))
        return {
            "initial_cash": float(config.get("initial_cash", p["initial_cash"])),
            "commission": float(config.get("commission", p["commission"])),
            "slippage": float(slip or 0.0),
            "leverage": float(config.get("leverage", p["leverage"])),
        }


class OandaBroker(PluginBase):
    """Live OANDA v20 broker — hard-gated stub (non-goal carried over from
    the reference, oanda_broker.py:43-46): refuses to construct unless
    GYMFX_ENABLE_LIVE=1."""

    plugin_params = {
        "initial_cash": 10000.0,
        "commission": 0.0,
        "slippage_perc": 0.0,
        "leverage": 1.0,
        "oanda_account_id": None,
        "oanda_token": None,
        "oanda_practice": True,
    }

    def broker_params(self, config: Dict[str, Any]) -> Dict[str, float]:
        if os.environ.get("GYMFX_ENABLE_LIVE") != "1":
            raise RuntimeError(
                "oanda_broker is live-trading and gated: set GYMFX_ENABLE_LIVE=1 "
                "to acknowledge. Simulated runs should use default_broker."
            )
        raise NotImplementedError(
            "live OANDA trading is a gated stub in this build (reference parity: "
            "oanda_broker.py is the same gate in front of bt.stores.OandaStore)."
        )
