"""Execution-cost-profile schema validation (reference semantics
simulation_engines/contracts.py:49-106; test idiom from
tests/test_simulation_engine_contracts.py)."""
import json
from decimal import Decimal

import pytest

from gymfx_amd.contracts import ExecutionCostProfile, load_execution_cost_profile

VALID = {
    "schema_version": "execution_cost_profile.v1",
    "profile_id": "test_v1",
    "commission_rate_per_side": "0.00002",
    "full_spread_rate": "0.0001",
    "slippage_bps_per_side": "0.5",
    "latency_ms": 5,
    "financing_enabled": False,
    "intrabar_collision_policy": "worst_case",
    "limit_fill_policy": "conservative",
    "margin_model": "leveraged",
    "enforce_margin_preflight": True,
    "random_seed": 7,
}


def test_valid_profile_roundtrip(tmp_path):
    path = tmp_path / "p.json"
    path.write_text(json.dumps(VALID))
    prof = load_execution_cost_profile(path)
    assert prof.profile_id == "test_v1"
    assert prof.slippage_rate_per_side == Decimal("0.00005")
    assert prof.quote_adverse_rate_per_side == Decimal("0.0001") / 2 + Decimal("0.00005")
    f = prof.as_floats()
    assert f["quote_adverse_rate_per_side"] == pytest.approx(0.0001)


def test_missing_field_rejected():
    raw = dict(VALID)
    del raw["margin_model"]
    with pytest.raises(ValueError, match="missing fields"):
        ExecutionCostProfile.from_dict(raw)


def test_bad_schema_version_rejected():
    raw = dict(VALID)
    raw["schema_version"] = "execution_cost_profile.v2"
    with pytest.raises(ValueError, match="schema_version"):
        ExecutionCostProfile.from_dict(raw)


@pytest.mark.parametrize(
    "field,value,msg",
    [
        ("commission_rate_per_side", "-1", "cannot be negative"),
        ("full_spread_rate", "1.5", "below 1"),
        ("latency_ms", -1, "latency_ms"),
        ("intrabar_collision_policy", "bogus", "intrabar_collision_policy"),
        ("limit_fill_policy", "bogus", "limit_fill_policy"),
        ("margin_model", "bogus", "margin_model"),
    ],
)
def test_invalid_values_rejected(field, value, msg):
    raw = dict(VALID)
    raw[field] = value
    with pytest.raises(ValueError, match=msg):
        ExecutionCostProfile.from_dict(raw)


def test_profile_overrides_env_costs(tmp_path):
    path = tmp_path / "p.json"
    path.write_text(json.dumps(VALID))
    from gymfx_amd.envs.params import EnvParams

    p = EnvParams.from_config({"execution_cost_profile": str(path)})
    assert p.commission == pytest.approx(0.00002)
    assert p.slippage == pytest.approx(0.0001)  # spread/2 + slippage
