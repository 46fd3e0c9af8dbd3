"""Metrics plugins: base digest + trading.metrics.v1 risk-adjusted fields
(reference semantics metrics_plugins/{default,trading}_metrics.py; idiom of
tests/test_trading_metrics.py)."""
import pytest

from gymfx_amd.plugins.metrics import DefaultMetrics, TradingMetrics

ANALYZERS = {
    "trades": {"total": {"total": 10}, "won": {"total": 6}, "lost": {"total": 4},
               "pnl": {"net": {"average": 1.5}}},
    "sharpe": {"sharperatio": 0.8},
    "drawdown": {"max": {"drawdown": 12.5, "moneydown": 125.0}},
    "sqn": {"sqn": 1.9},
}


def test_default_metrics_digest():
    s = DefaultMetrics().summarize(
        initial_cash=1000.0, final_equity=1100.0, analyzers=ANALYZERS, config={}
    )
    assert s["total_return"] == pytest.approx(0.1)
    assert s["max_drawdown_pct"] == 12.5
    assert s["trades_total"] == 10
    assert s["avg_trade_pnl"] == 1.5


def test_trading_metrics_rap():
    s = TradingMetrics().summarize(
        initial_cash=1000.0, final_equity=1100.0, analyzers=ANALYZERS,
        config={"risk_lambda": 2.0},
    )
    assert s["metric_schema"] == "trading.metrics.v1"
    assert s["max_drawdown_fraction"] == pytest.approx(0.125)
    assert s["rap"] == pytest.approx(0.1 - 2.0 * 0.125)
    assert "annual_return" not in s  # no evaluation_years -> no annualization


def test_trading_metrics_annualization():
    s = TradingMetrics().summarize(
        initial_cash=1000.0, final_equity=1210.0, analyzers={}, config={
            "evaluation_years": 2.0,
        },
    )
    assert s["annual_return"] == pytest.approx(0.1)
    assert s["annual_rap"] == pytest.approx(s["rap"] / 2.0)


def test_total_return_identity():
    s = DefaultMetrics().summarize(
        initial_cash=10000.0, final_equity=10123.0, analyzers={}, config={}
    )
    assert s["total_return"] == pytest.approx((10123.0 - 10000.0) / 10000.0)
