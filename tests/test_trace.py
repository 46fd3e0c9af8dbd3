"""PhaseTimer / TraceWriter unit behavior (utils/trace.py)."""
import json
import time

import torch

from gymfx_amd.utils.trace import PhaseTimer, TraceWriter


def test_phase_timer_accumulates_and_drains():
    t = PhaseTimer(torch.device("cpu"))
    with t.phase("a"):
        time.sleep(0.01)
    with t.phase("a"):
        time.sleep(0.01)
    with t.phase("b"):
        pass
    out = t.drain()
    assert out["a"] >= 15.0  # two ~10ms sleeps, accumulated
    assert out["b"] >= 0.0
    assert t.drain() == {}  # drained


def test_phase_timer_records_on_exception():
    t = PhaseTimer(torch.device("cpu"))
    try:
        with t.phase("x"):
            raise RuntimeError("boom")
    except RuntimeError:
        pass
    assert "x" in t.drain()


def test_trace_writer_appends_jsonl(tmp_path):
    p = tmp_path / "t.jsonl"
    w = TraceWriter(str(p))
    w.write({"u": 0})
    w.write({"u": 1})
    w.close()
    w2 = TraceWriter(str(p))  # append mode
    w2.write({"u": 2})
    w2.close()
    recs = [json.loads(l) for l in p.read_text().splitlines()]
    assert [r["u"] for r in recs] == [0, 1, 2]
