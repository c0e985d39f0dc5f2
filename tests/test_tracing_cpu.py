"""Tracing subsystem: chrome-trace span collection and
JSON export; dag.dot dump."""

import json


from tepdist_amd.runtime.task_graph import build_task_dag
from tepdist_amd.utils.tracing import Tracer


def test_tracer_spans_and_save(tmp_path):
    p = str(tmp_path / "trace.json")
    tr = Tracer(p)
    assert tr.enabled
    with tr.span("step", args={"i": 1}):
        with tr.span("fw"):
            pass
    tr.save()
    d = json.load(open(p))
    names = [e["name"] for e in d["traceEvents"]]
    assert "step" in names and "fw" in names
    assert all(e["ph"] == "X" and e["dur"] >= 0 for e in d["traceEvents"])


def test_tracer_disabled_without_env(monkeypatch):
    monkeypatch.delenv("TEPDIST_TRACE", raising=False)
    tr = Tracer()
    with tr.span("x"):
        pass
    assert not tr.events


def test_dag_dot_dump(tmp_path):
    dag = build_task_dag(2, 2)
    p = str(tmp_path / "dag.dot")
    dag.dump_dot(p)
    s = open(p).read()
    assert "digraph" in s and "->" in s
