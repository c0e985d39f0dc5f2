"""Native runtime core (runtime/csrc/rt_core.cpp) vs the Python fallback:
the C++ scheduler simulator, GC plan and dominance tree must agree with
the pure-Python implementations on the same DAGs (the reference implements
these in C++, SURVEY.md §2.6: TaskScheduler / lifetime tracker /
TaskDAG dominance)."""

import pytest

from tepdist_amd.runtime.scheduler import TaskScheduler, _rt
from tepdist_amd.runtime.task_graph import build_task_dag, idoms


pytestmark = pytest.mark.skipif(_rt is None, reason="native core not built")


def _dags():
    yield build_task_dag(1, 1)
    yield build_task_dag(1, 4)
    yield build_task_dag(2, 4)
    yield build_task_dag(4, 8, dp_degree=2)


def test_native_matches_python_schedule():
    for dag in _dags():
        for limit in (0, 2):
            py = TaskScheduler(dag, micro_num_limit=limit).schedule(
                native=False)
            nat = TaskScheduler(dag, micro_num_limit=limit).schedule(
                native=True)
            assert py.order == nat.order, (py.policy, nat.policy)
            assert abs(py.makespan - nat.makespan) < 1e-9


def test_native_gc_plan_releases_every_consumed_output():
    dag = build_task_dag(2, 4)
    sched = TaskScheduler(dag)
    res = sched.schedule()
    plan = sched.gc_plan(res.order)
    released = [p for v in plan.values() for p in v]
    producers = [t.id for t in dag.tasks.values() if t.children]
    assert sorted(released) == sorted(producers)
    # a buffer is only released at (or after) its last consumer
    pos = {}
    for lst in res.order.values():
        for i, t in enumerate(lst):
            pos[t] = i
    for releaser, prods in plan.items():
        for p in prods:
            for c in dag.tasks[p].children:
                assert pos.get(c, -1) <= pos[releaser]


def test_dominance_tree():
    dag = build_task_dag(2, 2)
    dom = idoms(dag)
    roots = [t.id for t in dag.tasks.values() if not t.parents]
    for r in roots:
        assert dom[r] == -1
    # a single-parent task is dominated by that parent
    for t in dag.tasks.values():
        if len(t.parents) == 1:
            assert dom[t.id] == t.parents[0]


def test_dominance_native_matches_python(monkeypatch):
    dag = build_task_dag(4, 4)
    nat = idoms(dag)
    import tepdist_amd.runtime.task_graph as tg
    real_import = __import__

    def no_rt(name, *a, **k):
        if name.endswith("_tepdist_rt"):
            raise ImportError(name)
        return real_import(name, *a, **k)

    import builtins
    monkeypatch.setattr(builtins, "__import__", no_rt)
    py = tg.idoms(dag)
    assert nat == py
