"""Metrics plugins: end-of-run summary from analyzer-equivalents.

Schema parity with /root/reference/metrics_plugins/default_metrics.py:22-60
and trading_metrics.py (schema id ``trading.metrics.v1``, rap fields,
optional annualization only when evaluation_years is given).
"""
from __future__ import annotations

import math
from typing import Any, Dict

from .base import PluginBase


def _get(d: Any, *path: str, default: Any = None) -> Any:
    cur: Any = d
    for k in path:
        if cur is None:
            return default
        if hasattr(cur, "get"):
            cur = cur.get(k, None)
        else:
            return default
    return cur if cur is not None else default


def _finite_or_zero(value: Any) -> float:
    try:
        result = float(value)
    except (TypeError, ValueError):
        return 0.0
    return result if math.isfinite(result) else 0.0


class DefaultMetrics(PluginBase):
    plugin_params: Dict[str, Any] = {}

    def summarize(
        self,
        *,
        initial_cash: float,
        final_equity: float,
        analyzers: Dict[str, Any],
        config: Dict[str, Any],
    ) -> Dict[str, Any]:
        trades = analyzers.get("trades") or {}
        sharpe = analyzers.get("sharpe") or {}
        drawdown = analyzers.get("drawdown") or {}
        sqn = analyzers.get("sqn") or {}
        total_return = (
            (float(final_equity) / float(initial_cash) - 1.0) if initial_cash else 0.0
        )
        return {
            "initial_cash": float(initial_cash),
            "final_equity": float(final_equity),
            "total_return": float(total_return),
            "max_drawdown_pct": _get(drawdown, "max", "drawdown"),
            "max_drawdown_money": _get(drawdown, "max", "moneydown"),
            "sharpe_ratio": _get(sharpe, "sharperatio"),
            "sqn": _get(sqn, "sqn"),
            "trades_total": _get(trades, "total", "total", default=0),
            "trades_won": _get(trades, "won", "total", default=0),
            "trades_lost": _get(trades, "lost", "total", default=0),
            "avg_trade_pnl": _get(trades, "pnl", "net", "average"),
        }


class TradingMetrics(DefaultMetrics):
    """Adds unit-safe risk-adjusted fields on top of the base summary."""

    plugin_params: Dict[str, Any] = {
        "risk_lambda": 1.0,
        "metric_schema": "trading.metrics.v1",
    }

    def summarize(
        self,
        *,
        initial_cash: float,
        final_equity: float,
        analyzers: Dict[str, Any],
        config: Dict[str, Any],
    ) -> Dict[str, Any]:
        summary = super().summarize(
            initial_cash=initial_cash,
            final_equity=final_equity,
            analyzers=analyzers,
            config=config,
        )
        drawdown_pct = _finite_or_zero(summary.get("max_drawdown_pct"))
        total_return = _finite_or_zero(summary.get("total_return"))
        risk_lambda = float(
            config.get(
                "risk_lambda",
                config.get("risk_penalty_lambda", self.params["risk_lambda"]),
            )
        )
        drawdown_fraction = max(0.0, drawdown_pct / 100.0)
        rap = total_return - risk_lambda * drawdown_fraction
        summary.update(
            {
                "metric_schema": str(
                    config.get("metric_schema", self.params["metric_schema"])
                ),
                "max_drawdown_fraction": drawdown_fraction,
                "risk_penalty_lambda": risk_lambda,
                "risk_adjusted_total_return": rap,
                "rap": rap,
            }
        )
        years = config.get("evaluation_years")
        if years is not None and float(years) > 0:
            summary["annual_return"] = (1.0 + total_return) ** (1.0 / float(years)) - 1.0
            summary["annual_rap"] = rap / float(years)
        return summary
