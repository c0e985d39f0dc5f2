// Native runtime core: the compile-time machinery the reference implements
// in C++ — TaskScheduler's event-driven list-scheduling
// simulator (task_scheduler.h:86-374), the output-buffer lifetime tracker
// that drives the GC plan (lifetime_tracker.h:32-86,
// execution_plan.cc:28-68), and the TaskDAG dominance tree (Cooper et al.,
// task_graph.h:643-717). Exposed via pybind11; runtime/scheduler.py and
// task_graph.py call into this and keep a pure-Python fallback.
//
// The simulator mirrors runtime/scheduler.py exactly (same priority
// tuples, same memory/micro-limit accounting) so Python and native
// schedules are bit-identical — tests assert that.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <cstdint>
#include <map>
#include <queue>
#include <tuple>
#include <unordered_map>
#include <vector>

namespace py = pybind11;

namespace {

struct SimResult {
  std::map<int64_t, std::vector<int64_t>> order;  // device -> task ids
  double makespan = 0.0;
  std::map<int64_t, double> peak;
  bool feasible = false;
};

// kind: 0 = backward compute, 1 = forward compute, -1 = everything else
SimResult simulate(const std::vector<int>& kind,
                   const std::vector<int64_t>& device,
                   const std::vector<int64_t>& micro,
                   const std::vector<double>& dur,
                   const std::vector<double>& out_bytes,
                   const std::vector<int64_t>& release_at,
                   const std::vector<int64_t>& parent_off,
                   const std::vector<int64_t>& parent_ids,
                   const std::vector<int64_t>& child_off,
                   const std::vector<int64_t>& child_ids, bool bw_first,
                   int64_t micro_limit, double mem_cap) {
  const int64_t n = (int64_t)kind.size();
  SimResult res;
  std::vector<int64_t> indeg(n);
  for (int64_t i = 0; i < n; ++i) indeg[i] = parent_off[i + 1] - parent_off[i];

  using Prio = std::tuple<int64_t, int64_t, int64_t>;  // (kind, micro, id)
  auto prio = [&](int64_t t) -> Prio {
    if (bw_first) return {kind[t] == 0 ? 0 : (kind[t] == 1 ? 1 : -1),
                          micro[t], t};
    return {t, 0, 0};
  };
  using QEntry = std::pair<Prio, int64_t>;
  std::map<int64_t, std::priority_queue<QEntry, std::vector<QEntry>,
                                        std::greater<QEntry>>> ready;
  std::unordered_map<int64_t, double> dev_free, mem;
  std::unordered_map<int64_t, int64_t> inflight_fw;
  std::vector<double> finish(n, 0.0);

  auto push = [&](int64_t t) { ready[device[t]].push({prio(t), t}); };
  for (int64_t i = 0; i < n; ++i)
    if (indeg[i] == 0) push(i);

  std::priority_queue<std::pair<double, int64_t>,
                      std::vector<std::pair<double, int64_t>>,
                      std::greater<std::pair<double, int64_t>>> events;
  double time_now = 0.0;
  int64_t scheduled = 0, guard = 0;
  while (scheduled < n) {
    if (++guard > 10 * n + 100) return res;  // infeasible
    bool progressed = false;
    for (auto& [dev, q] : ready) {
      while (!q.empty()) {
        const int64_t t = q.top().second;
        if (micro_limit > 0 && kind[t] == 1 &&
            inflight_fw[dev] >= micro_limit)
          break;
        q.pop();
        const double start = std::max(dev_free.count(dev) ? dev_free[dev]
                                                          : 0.0, time_now);
        const double end = start + dur[t];
        dev_free[dev] = end;
        finish[t] = end;
        res.order[dev].push_back(t);
        if (kind[t] == 1) {
          inflight_fw[dev] += 1;
          mem[dev] += out_bytes[t];
          auto it = res.peak.find(dev);
          if (it == res.peak.end() || mem[dev] > it->second)
            res.peak[dev] = mem[dev];
          if (mem[dev] > mem_cap) return res;  // infeasible
        }
        events.push({end, t});
        ++scheduled;
        progressed = true;
      }
    }
    if (!progressed) {
      if (events.empty()) return res;
      auto [tm, done] = events.top();
      events.pop();
      time_now = tm;
      if (kind[done] == 0) {
        const int64_t dev = device[done];
        inflight_fw[dev] = std::max<int64_t>(inflight_fw[dev] - 1, 0);
        mem[dev] = std::max(mem[dev] - out_bytes[done], 0.0);
        for (int64_t pi = parent_off[done]; pi < parent_off[done + 1]; ++pi) {
          const int64_t p = parent_ids[pi];
          if (release_at[p] == done)
            mem[dev] = std::max(mem[dev] - out_bytes[p], 0.0);
        }
      }
      for (int64_t ci = child_off[done]; ci < child_off[done + 1]; ++ci) {
        const int64_t c = child_ids[ci];
        if (--indeg[c] == 0) push(c);
      }
    }
  }
  res.makespan = *std::max_element(finish.begin(), finish.end());
  res.feasible = true;
  return res;
}

// GC plan: for every task (in its device's scheduled order), the list of
// PRODUCER tasks whose output buffer dies once this task completes — i.e.
// this task is the producer's last consumer in execution order. The
// reference computes this with static ref counts + the dominance tree
// (MakeTaskGraphGCPlan); with per-device total orders the last consumer
// position is exact.
std::map<int64_t, std::vector<int64_t>> gc_plan(
    const std::vector<int64_t>& child_off,
    const std::vector<int64_t>& child_ids,
    const std::map<int64_t, std::vector<int64_t>>& order) {
  // global completion index approximation: device order position
  std::unordered_map<int64_t, int64_t> pos;
  for (const auto& [dev, ids] : order) {
    int64_t p = 0;
    for (int64_t t : ids) pos[t] = p++;
  }
  std::map<int64_t, std::vector<int64_t>> plan;
  const int64_t n = (int64_t)child_off.size() - 1;
  for (int64_t prod = 0; prod < n; ++prod) {
    int64_t last = -1, best = -1;
    for (int64_t ci = child_off[prod]; ci < child_off[prod + 1]; ++ci) {
      const int64_t c = child_ids[ci];
      const int64_t p = pos.count(c) ? pos[c] : -1;
      if (p > best) {
        best = p;
        last = c;
      }
    }
    if (last >= 0) plan[last].push_back(prod);
  }
  return plan;
}

// Immediate dominators (Cooper, Harvey & Kennedy "A Simple, Fast Dominance
// Algorithm") over the task DAG, virtual root = tasks with no parents.
std::vector<int64_t> idom_tree(const std::vector<int64_t>& parent_off,
                               const std::vector<int64_t>& parent_ids) {
  const int64_t n = (int64_t)parent_off.size() - 1;
  // topological order via indegrees (DAG ids are not assumed presorted)
  std::vector<int64_t> indeg(n), topo;
  topo.reserve(n);
  std::vector<std::vector<int64_t>> children(n);
  for (int64_t i = 0; i < n; ++i) {
    indeg[i] = parent_off[i + 1] - parent_off[i];
    for (int64_t pi = parent_off[i]; pi < parent_off[i + 1]; ++pi)
      children[parent_ids[pi]].push_back(i);
  }
  std::vector<int64_t> stack;
  for (int64_t i = 0; i < n; ++i)
    if (indeg[i] == 0) stack.push_back(i);
  while (!stack.empty()) {
    const int64_t u = stack.back();
    stack.pop_back();
    topo.push_back(u);
    for (int64_t c : children[u])
      if (--indeg[c] == 0) stack.push_back(c);
  }
  std::vector<int64_t> rpo_idx(n, -1);
  for (int64_t i = 0; i < (int64_t)topo.size(); ++i) rpo_idx[topo[i]] = i;

  std::vector<int64_t> idom(n, -2);  // -2 undefined, -1 virtual root
  auto intersect = [&](int64_t a, int64_t b) {
    while (a != b) {
      if (a == -1 || b == -1) return (int64_t)-1;
      while (a != -1 && rpo_idx[a] > rpo_idx[b]) a = idom[a];
      while (b != -1 && a != -1 && rpo_idx[b] > rpo_idx[a]) b = idom[b];
      if (a == -1 || b == -1) return (int64_t)-1;
    }
    return a;
  };
  bool changed = true;
  while (changed) {
    changed = false;
    for (int64_t u : topo) {
      int64_t nd = -2;
      for (int64_t pi = parent_off[u]; pi < parent_off[u + 1]; ++pi) {
        const int64_t p = parent_ids[pi];
        if (idom[p] == -2 && parent_off[p + 1] - parent_off[p] > 0)
          continue;  // unprocessed
        nd = (nd == -2) ? p : intersect(nd, p);
      }
      if (nd == -2) nd = -1;  // entry task: dominated by virtual root
      if (idom[u] != nd) {
        idom[u] = nd;
        changed = true;
      }
    }
  }
  return idom;
}

}  // namespace

PYBIND11_MODULE(_tepdist_rt, m) {
  m.doc() = "tepdist native runtime core (scheduler sim, GC, dominance)";
  py::class_<SimResult>(m, "SimResult")
      .def_readonly("order", &SimResult::order)
      .def_readonly("makespan", &SimResult::makespan)
      .def_readonly("peak", &SimResult::peak)
      .def_readonly("feasible", &SimResult::feasible);
  m.def("simulate", &simulate, py::arg("kind"), py::arg("device"),
        py::arg("micro"), py::arg("dur"), py::arg("out_bytes"),
        py::arg("release_at"), py::arg("parent_off"), py::arg("parent_ids"),
        py::arg("child_off"), py::arg("child_ids"), py::arg("bw_first"),
        py::arg("micro_limit"), py::arg("mem_cap"));
  m.def("gc_plan", &gc_plan, py::arg("child_off"), py::arg("child_ids"),
        py::arg("order"));
  m.def("idom_tree", &idom_tree, py::arg("parent_off"),
        py::arg("parent_ids"));
}
