import json

from bigclam.utils.metrics import MetricsLogger, PhaseTimer


def test_metrics_rank_gating(tmp_path, capsys):
    p0 = tmp_path / "r0.jsonl"
    p1 = tmp_path / "r1.jsonl"
    m0 = MetricsLogger(str(p0), rank=0, quiet=True)
    m1 = MetricsLogger(str(p1), rank=1, quiet=False)
    m0.log({"sweep": 1, "llh": -2.0})
    m1.log({"sweep": 1, "llh": -2.0})
    m0.close()
    m1.close()
    lines = [json.loads(l) for l in open(p0)]
    assert len(lines) == 1 and lines[0]["llh"] == -2.0 and "ts" in lines[0]
    assert not p1.exists()  # non-zero ranks write (and print) nothing
    assert capsys.readouterr().err == ""


def test_phase_timer_accumulates():
    t = PhaseTimer(sync=False)
    with t.phase("a"):
        pass
    with t.phase("b"):
        pass
    with t.phase("a"):
        pass
    assert set(t.times) == {"a", "b"}
    assert t.times["a"] >= 0.0 and t.times["b"] >= 0.0
    # stop() without start is a no-op
    t.stop()
