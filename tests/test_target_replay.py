"""TargetReplay engine: intrabar execution paths, collision/limit policies,
latency, fill model, margin models, execution reports.

Mirrors the reference's bakeoff fixtures (flash-crash collision,
margin rejection, financing — /root/reference/simulation_engines/bakeoff.py:116-210)
against the MI355X-native target-position replay (gymfx_amd/target_replay.py).
"""
from decimal import Decimal

import pytest

from gymfx_amd.contracts import (ExecutionCostProfile, InstrumentSpec,
                                 MarketFrame, TargetAction)
from gymfx_amd.target_replay import (FillModel, TargetReplay,
                                     export_execution_reports)

NS_MIN = 60 * 1_000_000_000
BASE_NS = 1_704_153_600 * 1_000_000_000  # 2024-01-02T00:00:00Z


def _ts(i: int) -> int:
    return BASE_NS + i * NS_MIN


def _profile(**over):
    raw = {
        "schema_version": "execution_cost_profile.v1", "profile_id": "test",
        "commission_rate_per_side": "0.00002", "full_spread_rate": "0",
        "slippage_bps_per_side": "0", "latency_ms": 0,
        "financing_enabled": False,
        "intrabar_collision_policy": "worst_case",
        "limit_fill_policy": "touch", "margin_model": "leveraged",
        "enforce_margin_preflight": False, "random_seed": 7,
    }
    raw.update(over)
    return ExecutionCostProfile.from_dict(raw)


EURUSD = InstrumentSpec(
    symbol="EUR/USD", venue="SIM", base_currency="EUR", quote_currency="USD",
    price_precision=5, size_precision=0,
    margin_init=Decimal("0.03"), margin_maint=Decimal("0.03"))


def _frame(i, o, h, l, c, path=None):
    d = Decimal
    return MarketFrame("EUR/USD.SIM", 1, _ts(i), d(o), d(h), d(l), d(c),
                       Decimal("1000000"),
                       execution_path=tuple(d(p) for p in path) if path else None)


def _flat_frames(n, px="1.10000", start=0):
    return [_frame(start + i, px, px, px, px) for i in range(n)]


def _long_with_brackets(frames, profile, sl="1.09800", tp="1.10200",
                        units="1000", cash="100000"):
    actions = [TargetAction("EUR/USD.SIM", _ts(1), Decimal(units), "a1",
                            stop_loss_price=Decimal(sl),
                            take_profit_price=Decimal(tp))]
    return TargetReplay(profile).run(
        instrument_specs=[EURUSD], frames=frames, actions=actions,
        initial_cash=Decimal(cash))


def _fills(result, kind=None):
    return [e for e in result["events"] if e["event_type"] == "order_filled"
            and (kind is None or e["kind"] == kind)]


# ---------------------------------------------------------------------------
# intrabar collision: the flash-crash fixture (ref bakeoff.py:116-163)
# ---------------------------------------------------------------------------

def test_execution_path_dip_first_hits_stop():
    """Path dips through the SL before rallying through the TP: the stop
    must fill (at the dip quote), the take-profit must not."""
    frames = [
        _frame(1, "1.10000", "1.10010", "1.09990", "1.10000"),
        _frame(2, "1.10000", "1.10300", "1.09700", "1.10200",
               path=["1.10000", "1.09700", "1.10300", "1.10200"]),
        _frame(3, "1.10200", "1.10210", "1.10190", "1.10200"),
    ]
    r = _long_with_brackets(frames, _profile())
    entry = _fills(r, "order_filled")
    sl = _fills(r, "bracket_sl_fill")
    assert len(entry) == 1 and entry[0]["price"] == "1.10000"
    assert len(sl) == 1 and sl[0]["price"] == "1.09700"
    assert not _fills(r, "bracket_tp_fill")
    assert r["positions"].get("EUR/USD.SIM", {}).get("units", "0") == "0"


def test_execution_path_rally_first_hits_take_profit():
    """Same bar extremes, opposite path order: the TP fills, the SL does
    not — path order IS the collision outcome."""
    frames = [
        _frame(1, "1.10000", "1.10010", "1.09990", "1.10000"),
        _frame(2, "1.10000", "1.10300", "1.09700", "1.10200",
               path=["1.10000", "1.10300", "1.09700", "1.10200"]),
        _frame(3, "1.10200", "1.10210", "1.10190", "1.10200"),
    ]
    r = _long_with_brackets(frames, _profile())
    assert len(_fills(r, "bracket_tp_fill")) == 1
    assert not _fills(r, "bracket_sl_fill")
    # limit fills at the limit price (no gap at that point)
    assert _fills(r, "bracket_tp_fill")[0]["price"] == "1.10200"


@pytest.mark.parametrize("policy,expect_kind", [
    ("worst_case", "bracket_sl_fill"),   # adverse extreme first for a long
    ("ohlc", "bracket_tp_fill"),         # high printed before low
    ("adaptive", "bracket_sl_fill"),     # up bar -> low assumed first
])
def test_synthetic_collision_policies(policy, expect_kind):
    """No execution_path: the collision policy decides the synthetic point
    order when both SL and TP lie inside one bar."""
    frames = [
        _frame(1, "1.10000", "1.10010", "1.09990", "1.10000"),
        _frame(2, "1.10000", "1.10300", "1.09700", "1.10250"),  # up bar
        _frame(3, "1.10250", "1.10260", "1.10240", "1.10250"),
    ]
    r = _long_with_brackets(frames, _profile(intrabar_collision_policy=policy))
    hits = [e["kind"] for e in _fills(r) if e["kind"].startswith("bracket")]
    assert hits == [expect_kind]


def test_adaptive_down_bar_checks_high_first():
    frames = [
        _frame(1, "1.10000", "1.10010", "1.09990", "1.10000"),
        _frame(2, "1.10000", "1.10300", "1.09700", "1.09750"),  # down bar
        _frame(3, "1.09750", "1.09760", "1.09740", "1.09750"),
    ]
    r = _long_with_brackets(frames,
                            _profile(intrabar_collision_policy="adaptive"))
    hits = [e["kind"] for e in _fills(r) if e["kind"].startswith("bracket")]
    assert hits == ["bracket_tp_fill"]


# ---------------------------------------------------------------------------
# limit_fill_policy
# ---------------------------------------------------------------------------

def _tp_touch_frames(high):
    return [
        _frame(1, "1.10000", "1.10010", "1.09990", "1.10000"),
        _frame(2, "1.10000", high, "1.09990", "1.10000"),
        _frame(3, "1.10000", "1.10010", "1.09990", "1.10000"),
    ]


def test_limit_policy_touch_fills_at_touch():
    r = _long_with_brackets(_tp_touch_frames("1.10200"),
                            _profile(limit_fill_policy="touch"),
                            sl="1.09000", tp="1.10200")
    assert len(_fills(r, "bracket_tp_fill")) == 1


def test_limit_policy_cross_requires_strict_cross():
    prof = _profile(limit_fill_policy="cross")
    touch_only = _long_with_brackets(_tp_touch_frames("1.10200"), prof,
                                     sl="1.09000", tp="1.10200")
    assert not _fills(touch_only, "bracket_tp_fill")
    crossed = _long_with_brackets(_tp_touch_frames("1.10201"), prof,
                                  sl="1.09000", tp="1.10200")
    assert len(_fills(crossed, "bracket_tp_fill")) == 1


def test_limit_policy_conservative_no_gap_improvement():
    """Bar 3 gaps open above the TP: touch/cross fill at the (better) open
    quote, conservative fills at the limit price exactly."""
    frames = [
        _frame(1, "1.10000", "1.10010", "1.09990", "1.10000"),
        _frame(2, "1.10000", "1.10010", "1.09990", "1.10000"),
        _frame(3, "1.10500", "1.10510", "1.10490", "1.10500"),  # gap up
        _frame(4, "1.10500", "1.10510", "1.10490", "1.10500"),
    ]
    gap = _long_with_brackets(frames, _profile(limit_fill_policy="touch"),
                              sl="1.09000", tp="1.10200")
    cons = _long_with_brackets(frames,
                               _profile(limit_fill_policy="conservative"),
                               sl="1.09000", tp="1.10200")
    assert _fills(gap, "bracket_tp_fill")[0]["price"] == "1.10500"
    assert _fills(cons, "bracket_tp_fill")[0]["price"] == "1.10200"


# ---------------------------------------------------------------------------
# latency
# ---------------------------------------------------------------------------

def test_latency_delays_fill_by_bars():
    frames = _flat_frames(6)
    act = [TargetAction("EUR/USD.SIM", _ts(1), Decimal("1000"), "a1")]
    fast = TargetReplay(_profile(latency_ms=0)).run(
        instrument_specs=[EURUSD], frames=frames, actions=act)
    slow = TargetReplay(_profile(latency_ms=90_000)).run(
        instrument_specs=[EURUSD], frames=frames, actions=act)
    assert _fills(fast)[0]["ts_event_ns"] == _ts(2)
    # 90 s latency on 1-minute bars: executable only after ts+90s -> bar 3
    assert _fills(slow)[0]["ts_event_ns"] == _ts(3)


# ---------------------------------------------------------------------------
# margin models + preflight (ref bakeoff.py:164-177, nautilus_gym.py:129-143)
# ---------------------------------------------------------------------------

def test_margin_preflight_denies_oversized_target():
    frames = _flat_frames(4)
    act = [TargetAction("EUR/USD.SIM", _ts(1), Decimal("10000000"), "big")]
    r = TargetReplay(_profile(enforce_margin_preflight=True)).run(
        instrument_specs=[EURUSD], frames=frames, actions=act,
        initial_cash=Decimal("1000"))
    denied = [e for e in r["events"] if e["event_type"] == "order_denied"]
    assert len(denied) == 1 and denied[0]["action_id"] == "big"
    assert not _fills(r)
    assert r["final_balance"] == "1000"


def test_margin_model_standard_vs_leveraged():
    """standard holds notional x margin_init; leveraged divides by account
    leverage — the same order passes preflight only under leveraged."""
    frames = _flat_frames(4)
    # notional = 100k x 1.1 = 110k; standard margin = 3300 > 2000 cash;
    # leveraged margin = 3300/20 = 165 < 2000
    act = [TargetAction("EUR/USD.SIM", _ts(1), Decimal("100000"), "a1")]
    std = TargetReplay(_profile(margin_model="standard",
                                enforce_margin_preflight=True)).run(
        instrument_specs=[EURUSD], frames=frames, actions=act,
        initial_cash=Decimal("2000"))
    lev = TargetReplay(_profile(margin_model="leveraged",
                                enforce_margin_preflight=True)).run(
        instrument_specs=[EURUSD], frames=frames, actions=act,
        initial_cash=Decimal("2000"))
    assert any(e["event_type"] == "order_denied" for e in std["events"])
    assert len(_fills(lev)) == 1


# ---------------------------------------------------------------------------
# fill model + fees + determinism
# ---------------------------------------------------------------------------

def test_fill_model_prob_zero_stop_never_fills():
    frames = [
        _frame(1, "1.10000", "1.10010", "1.09990", "1.10000"),
        _frame(2, "1.10000", "1.10010", "1.09500", "1.09600"),
        _frame(3, "1.09600", "1.09610", "1.09590", "1.09600"),
    ]
    prof = _profile()
    fm = FillModel(random_seed=7, prob_fill_on_stop=0.0)
    r = TargetReplay(prof, fill_model=fm).run(
        instrument_specs=[EURUSD], frames=frames,
        actions=[TargetAction("EUR/USD.SIM", _ts(1), Decimal("1000"), "a1",
                              stop_loss_price=Decimal("1.09800"))])
    assert not _fills(r, "bracket_sl_fill")
    assert r["positions"]["EUR/USD.SIM"]["units"] == "1000"


def test_maker_taker_liquidity_flags():
    frames = [
        _frame(1, "1.10000", "1.10010", "1.09990", "1.10000"),
        _frame(2, "1.10000", "1.10300", "1.09990", "1.10250"),
        _frame(3, "1.10250", "1.10260", "1.10240", "1.10250"),
    ]
    r = _long_with_brackets(frames, _profile(intrabar_collision_policy="ohlc"),
                            sl="1.09000", tp="1.10200")
    entry = _fills(r, "order_filled")[0]
    tp = _fills(r, "bracket_tp_fill")[0]
    assert entry["liquidity"] == "taker"
    assert tp["liquidity"] == "maker"


def test_replay_is_deterministic_and_hashed():
    frames = [
        _frame(1, "1.10000", "1.10010", "1.09990", "1.10000"),
        _frame(2, "1.10000", "1.10300", "1.09700", "1.10200",
               path=["1.10000", "1.09700", "1.10300", "1.10200"]),
        _frame(3, "1.10200", "1.10210", "1.10190", "1.10200"),
    ]
    r1 = _long_with_brackets(frames, _profile())
    r2 = _long_with_brackets(frames, _profile())
    assert r1["result_hash"] == r2["result_hash"]
    assert r1["event_hash"] == r2["event_hash"]


# ---------------------------------------------------------------------------
# delta-to-target semantics (ref nautilus_gym.py:118-127)
# ---------------------------------------------------------------------------

def test_target_units_are_absolute_targets_not_deltas():
    frames = _flat_frames(6)
    acts = [
        TargetAction("EUR/USD.SIM", _ts(1), Decimal("1000"), "go-long"),
        TargetAction("EUR/USD.SIM", _ts(3), Decimal("-500"), "flip-short"),
    ]
    r = TargetReplay(_profile()).run(
        instrument_specs=[EURUSD], frames=frames, actions=acts)
    reqs = [e for e in r["events"] if e["event_type"] == "target_requested"]
    assert [e["delta_units"] for e in reqs] == ["1000", "-1500"]
    fills = _fills(r)
    assert [f["side"] for f in fills] == ["BUY", "SELL"]
    assert [f["quantity"] for f in fills] == ["1000", "1500"]
    assert r["positions"]["EUR/USD.SIM"]["units"] == "-500"


# ---------------------------------------------------------------------------
# financing (ref bakeoff.py:179-210)
# ---------------------------------------------------------------------------

def test_financing_rollover_event():
    import pandas as pd

    rate_data = pd.DataFrame([
        {"LOCATION": "EA19", "TIME": "2024-01", "Value": 5.0},
        {"LOCATION": "USA", "TIME": "2024-01", "Value": 4.0},
    ])
    # 21:58, 22:01, 22:02 UTC — position held across the 22:00 rollover
    base = 1_704_232_680  # 2024-01-02T21:58:00Z
    def f(sec):
        px = Decimal("1.10000")
        return MarketFrame("EUR/USD.SIM", 1, sec * 1_000_000_000, px,
                           px + Decimal("0.0001"), px - Decimal("0.0001"),
                           px, Decimal("1000000"))
    frames = [f(base), f(base + 180), f(base + 240)]
    acts = [TargetAction("EUR/USD.SIM", frames[0].ts_event_ns,
                         Decimal("1000"), "overnight-open"),
            TargetAction("EUR/USD.SIM", frames[2].ts_event_ns,
                         Decimal("0"), "overnight-close")]
    r = TargetReplay(_profile(financing_enabled=True)).run(
        instrument_specs=[EURUSD], frames=frames, actions=acts,
        financing_rate_data=rate_data)
    fin = [e for e in r["events"] if e["event_type"] == "financing"]
    assert len(fin) == 1
    # long EUR vs USD with EUR rate above USD rate -> positive carry
    assert Decimal(fin[0]["amount"]) > 0


def test_financing_requires_rate_data():
    with pytest.raises(ValueError, match="financing_rate_data"):
        TargetReplay(_profile(financing_enabled=True)).run(
            instrument_specs=[EURUSD], frames=_flat_frames(3), actions=[])


# ---------------------------------------------------------------------------
# execution-report export (ref bakeoff.py:306-374)
# ---------------------------------------------------------------------------

def test_execution_report_export():
    prof = _profile(full_spread_rate="0.0001", slippage_bps_per_side="0.05")
    frames = _flat_frames(4)
    acts = [TargetAction("EUR/USD.SIM", _ts(1), Decimal("1000"), "a1")]
    r = TargetReplay(prof).run(instrument_specs=[EURUSD], frames=frames,
                               actions=acts)
    reports = export_execution_reports(r, [EURUSD], prof)
    assert len(reports) == 1
    rep = reports[0]
    assert rep["schema"] == "gymfx.execution_report.v1"
    assert rep["trace_id"] == r["result_hash"]
    assert rep["order_intent_id"] == "a1"
    assert rep["requested_units"] == 1000.0
    assert rep["filled_units"] == 1000.0
    assert rep["state"] == "filled"
    assert rep["latency_ms"] == 0.0
    assert rep["spread_cost"] == pytest.approx(1000 * 1.1 * 0.0001 / 2)
    assert rep["commission"] > 0
    for key in ("object_id", "as_of", "producer", "requested_price",
                "filled_price", "slippage_cost", "financing",
                "conversion_cost", "broker_ids"):
        assert key in rep


def test_replay_cli_runs_flash_crash_example(tmp_path):
    """tools/run_target_replay.py end-to-end over the shipped example:
    the flash-crash path fills the STOP (dip printed first) and exports
    canonical execution reports."""
    import json
    import subprocess
    import sys
    from pathlib import Path

    repo = Path(__file__).resolve().parents[1]
    out = tmp_path / "result.json"
    reports = tmp_path / "reports.json"
    r = subprocess.run(
        [sys.executable, str(repo / "tools" / "run_target_replay.py"),
         str(repo / "examples" / "replay" / "flash_crash.json"),
         "--out", str(out), "--reports", str(reports)],
        capture_output=True, text=True, timeout=120)
    assert r.returncode == 0, r.stderr[-2000:]
    result = json.loads(out.read_text())
    kinds = [e.get("kind") for e in result["events"]
             if e["event_type"] == "order_filled"]
    assert kinds == ["order_filled", "bracket_sl_fill"]
    assert result["positions"]["EUR/USD.SIM"]["units"] == "0"
    reps = json.loads(reports.read_text())
    assert len(reps) == 2 and all(
        x["schema"] == "gymfx.execution_report.v1" for x in reps)


def _parallel_replay_worker(q):
    """Isolated-process evaluator (ref tools/nautilus_parallel_smoke.py):
    same inputs must hash identically across spawned processes."""
    import json
    from pathlib import Path

    from gymfx_amd.target_replay import TargetReplay
    import tools.run_target_replay as cli

    repo = Path(__file__).resolve().parents[1]
    raw = json.loads((repo / "examples" / "replay" / "flash_crash.json")
                     .read_text())
    profile, specs, frames, actions = cli.load_inputs(raw)
    result = TargetReplay(profile).run(instrument_specs=specs, frames=frames,
                                       actions=actions)
    q.put(result["result_hash"])


def test_parallel_evaluators_hash_identically():
    import multiprocessing as mp

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_parallel_replay_worker, args=(q,))
             for _ in range(2)]
    for p in procs:
        p.start()
    hashes = [q.get(timeout=120) for _ in procs]
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert hashes[0] == hashes[1]


def test_cross_engine_bakeoff_reconciles():
    """tools/engine_bakeoff.py: the vectorized engine, the scalar ledger
    and the Decimal target replay agree on the same action script
    (the reference's cross-engine bakeoff, natively)."""
    import sys
    from pathlib import Path

    sys.path.insert(0, str(Path(__file__).resolve().parents[1]))
    from tools.engine_bakeoff import run_bakeoff

    out = run_bakeoff(rows=250, steps=80, seed=13)
    assert out["vec_vs_ledger_ok"]
    assert out["vec_vs_target_replay_ok"], out


def test_event_stream_reconstructs_final_balance_property():
    """Independent reconciliation from IMMUTABLE fill facts (the
    reference's reconcile_fills idiom, bakeoff.py:228-303): rebuild the
    account from the event log alone — average-price netting over
    order_filled quantities/prices/commissions plus financing amounts —
    and it must land on the engine's final_balance for random walks and
    random target scripts."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    @settings(max_examples=30, deadline=None)
    @given(st.integers(min_value=0, max_value=2 ** 31 - 1),
           st.lists(st.sampled_from([-2000, -1000, 0, 1000, 2000]),
                    min_size=2, max_size=12))
    def check(seed, targets):
        import numpy as np

        rng = np.random.default_rng(seed)
        n = len(targets) + 4
        mids = 1.1 + np.cumsum(rng.normal(0, 5e-4, size=n))
        frames = []
        for i, m in enumerate(mids):
            o = m
            h = m + abs(rng.normal(0, 4e-4))
            l = m - abs(rng.normal(0, 4e-4))
            c = min(max(m + rng.normal(0, 2e-4), l), h)
            frames.append(_frame(i, f"{o:.5f}", f"{h:.5f}", f"{l:.5f}",
                                 f"{c:.5f}"))
        actions = [TargetAction("EUR/USD.SIM", _ts(i + 1),
                                Decimal(t), f"a{i}")
                   for i, t in enumerate(targets)]
        prof = _profile(commission_rate_per_side="0.00002",
                        slippage_bps_per_side="0.1",
                        random_seed=seed % 1000)
        res = TargetReplay(prof).run(
            instrument_specs=[EURUSD], frames=frames, actions=actions,
            initial_cash=Decimal("100000"))

        # -- rebuild from events only (conv = 1: USD quote) --------------
        bal = Decimal("100000")
        units = Decimal(0)
        avg = Decimal(0)
        for ev in res["events"]:
            if ev["event_type"] == "financing":
                bal += Decimal(ev["amount"])
            if ev["event_type"] != "order_filled":
                continue
            qty = Decimal(ev["quantity"])
            px = Decimal(ev["price"])
            signed = qty if ev["side"] == "BUY" else -qty
            bal -= Decimal(ev["commission"])
            if units != 0 and units * signed < 0:
                closing = min(abs(units), qty)
                bal += closing * (px - avg) * (1 if units > 0 else -1)
                new_units = units + signed
                if new_units == 0:
                    avg = Decimal(0)
                elif units * new_units < 0:
                    avg = px
            else:
                new_units = units + signed
                if units == 0:
                    avg = px
                elif new_units != 0:
                    avg = (abs(units) * avg + qty * px) / abs(new_units)
            units = new_units
        assert abs(bal - Decimal(res["final_balance"])) <= Decimal("1e-6"), (
            seed, targets, str(bal), res["final_balance"])

    check()


def test_future_market_mutation_cannot_change_earlier_fill_facts():
    """No-future-leakage at the engine level (reference
    test_nautilus_bakeoff.py:124-156): scaling the LAST bar's prices x5
    must leave every event with an earlier timestamp byte-identical."""
    import numpy as np
    from dataclasses import replace

    rng = np.random.default_rng(17)
    n = 20
    mids = 1.1 + np.cumsum(rng.normal(0, 5e-4, size=n))
    frames = [_frame(i, f"{m:.5f}", f"{m + 3e-4:.5f}", f"{m - 3e-4:.5f}",
                     f"{m:.5f}") for i, m in enumerate(mids)]
    actions = [TargetAction("EUR/USD.SIM", _ts(i + 1),
                            Decimal([1000, -1000, 0, 2000][i % 4]), f"a{i}")
               for i in range(12)]
    prof = _profile(commission_rate_per_side="0.00002",
                    slippage_bps_per_side="0.1")

    def run(fr):
        return TargetReplay(prof).run(
            instrument_specs=[EURUSD], frames=fr, actions=actions,
            initial_cash=Decimal("100000"))

    cutoff = max(f.ts_event_ns for f in frames)
    base = run(frames)
    mutated_frames = [
        replace(f, open=f.open * 5, high=f.high * 5, low=f.low * 5,
                close=f.close * 5) if f.ts_event_ns == cutoff else f
        for f in frames]
    mut = run(mutated_frames)
    base_prefix = [e for e in base["events"] if e["ts_event_ns"] < cutoff]
    mut_prefix = [e for e in mut["events"] if e["ts_event_ns"] < cutoff]
    assert base_prefix == mut_prefix and len(base_prefix) > 4
