// _sched — C++ gang-placement core (pybind11, CPU).
//
// Native implementation of grove_amd/scheduler/placement.py (same algorithm, same
// results — cross-checked by tests/test_placement_native.py). This is the hot path of
// the control plane at scale: one call per pending PodGang per scheduling pass, over
// potentially hundreds of nodes. Filter = capacity (cpu/mem/gpu/pods); Score = xGMI
// packing (one 8×MI355X hive ≈153 GB/s per-GPU ring bandwidth; cross-node placements
// are NIC-bound and score low).
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <cstdint>
#include <optional>
#include <string>
#include <tuple>
#include <vector>

namespace py = pybind11;

namespace {

constexpr double kXgmiLinkGBps = 153.0;
constexpr double kXgmiPeerLinks = 7.0;
constexpr double kNicGBps = 50.0;

struct Node {
  std::string name;
  int64_t cpu_milli;
  double mem_bytes;
  std::vector<int> gpu_ids;
  int64_t pods;
  // measured min per-link xGMI bandwidth of this pool's fabric (GB/s); the
  // topology agent feeds it from rsmi_minmax_bandwidth_get, default nominal
  double link_gbps = kXgmiLinkGBps;
};

struct Pod {
  std::string name;
  int64_t cpu_milli;
  double mem_bytes;
  int gpus;
};

bool fits(const Node& n, const Pod& p) {
  return n.cpu_milli >= p.cpu_milli && n.mem_bytes >= p.mem_bytes &&
         (int)n.gpu_ids.size() >= p.gpus && n.pods >= 1;
}

std::vector<int> take(Node& n, const Pod& p) {
  n.cpu_milli -= p.cpu_milli;
  n.mem_bytes -= p.mem_bytes;
  n.pods -= 1;
  std::vector<int> out(n.gpu_ids.begin(), n.gpu_ids.begin() + p.gpus);
  n.gpu_ids.erase(n.gpu_ids.begin(), n.gpu_ids.begin() + p.gpus);
  return out;
}

double placement_score(size_t n_nodes_used, int total_gpus,
                       double link_gbps = kXgmiLinkGBps) {
  if (total_gpus <= 1) return link_gbps * kXgmiPeerLinks;
  if (n_nodes_used <= 1) return link_gbps;
  return kNicGBps / (2.0 * (double)(n_nodes_used - 1));
}

using Assignment = std::tuple<std::string, std::string, std::vector<int>>;
using NodeState =
    std::tuple<std::string, int64_t, double, std::vector<int>, int64_t, double>;

// Returns (assignments, score, consumed-node-states) or None. The solver runs
// WITHOUT the GIL (a plain-C++ result is computed in a released scope; Python objects
// are built only after the GIL is re-acquired), so large placements overlap the
// Python controllers instead of serializing behind them.
struct SolveResult {
  bool feasible = false;
  std::vector<Assignment> assignments;
  double score = 0.0;
  std::vector<NodeState> consumed;
};

SolveResult solve(std::vector<Node>& nodes, std::vector<Pod>& pods) {
  SolveResult out;
  int total_gpus = 0;
  for (auto& p : pods) total_gpus += p.gpus;
  std::vector<size_t> order(pods.size());
  for (size_t i = 0; i < order.size(); ++i) order[i] = i;
  std::stable_sort(order.begin(), order.end(), [&](size_t a, size_t b) {
    if (pods[a].gpus != pods[b].gpus) return pods[a].gpus > pods[b].gpus;
    return pods[a].cpu_milli > pods[b].cpu_milli;
  });

  auto emit = [&](std::vector<Node>& state, double score) {
    out.feasible = true;
    out.score = score;
    out.consumed.reserve(state.size());
    for (auto& n : state)
      out.consumed.emplace_back(n.name, n.cpu_milli, n.mem_bytes, n.gpu_ids, n.pods,
                                n.link_gbps);
  };

  // Phase 1: single-pool best-fit (leaves fewest free GPUs behind; ties prefer
  // the pool with the higher measured link bandwidth — placement.py parity).
  int best = -1;
  size_t best_left = SIZE_MAX;
  double best_bw = -1.0;
  for (size_t i = 0; i < nodes.size(); ++i) {
    Node trial = nodes[i];
    bool ok = true;
    for (size_t oi : order) {
      if (!fits(trial, pods[oi])) { ok = false; break; }
      take(trial, pods[oi]);
    }
    if (ok && (trial.gpu_ids.size() < best_left ||
               (trial.gpu_ids.size() == best_left && nodes[i].link_gbps > best_bw))) {
      best = (int)i;
      best_left = trial.gpu_ids.size();
      best_bw = nodes[i].link_gbps;
    }
  }
  if (best >= 0) {
    for (size_t oi : order) {
      auto ids = take(nodes[best], pods[oi]);
      out.assignments.emplace_back(pods[oi].name, nodes[best].name, ids);
    }
    emit(nodes, placement_score(1, total_gpus, nodes[best].link_gbps));
    return out;
  }

  // Phase 2: minimal spread, first-fit-decreasing preferring already-used nodes with
  // the most free GPUs.
  std::vector<Node> state = nodes;  // work on a copy; rollback = infeasible return
  std::vector<char> used(state.size(), 0);
  size_t used_count = 0;
  for (size_t oi : order) {
    int pick = -1;
    for (int pass = 0; pass < 2 && pick < 0; ++pass) {
      // pass 0: only already-used nodes; pass 1: any node (largest free GPUs first)
      int best_i = -1;
      long best_key = -1;
      for (size_t i = 0; i < state.size(); ++i) {
        if (pass == 0 && !used[i]) continue;
        if (!fits(state[i], pods[oi])) continue;
        long key = (long)state[i].gpu_ids.size() * 1000000 + state[i].cpu_milli / 1000;
        if (key > best_key) { best_key = key; best_i = (int)i; }
      }
      pick = best_i;
    }
    if (pick < 0) return out;  // infeasible
    auto ids = take(state[pick], pods[oi]);
    out.assignments.emplace_back(pods[oi].name, state[pick].name, ids);
    if (!used[pick]) { used[pick] = 1; ++used_count; }
  }
  double min_link = kXgmiLinkGBps;
  bool any = false;
  for (size_t i = 0; i < state.size(); ++i)
    if (used[i]) { min_link = any ? std::min(min_link, state[i].link_gbps)
                                  : state[i].link_gbps; any = true; }
  emit(state, placement_score(used_count, total_gpus, min_link));
  return out;
}

py::object place_gang(std::vector<NodeState> node_states,
                      std::vector<std::tuple<std::string, int64_t, double, int>> pod_specs) {
  SolveResult res;
  {
    py::gil_scoped_release released;
    std::vector<Node> nodes;
    nodes.reserve(node_states.size());
    for (auto& t : node_states)
      nodes.push_back(Node{std::get<0>(t), std::get<1>(t), std::get<2>(t),
                           std::get<3>(t), std::get<4>(t), std::get<5>(t)});
    std::vector<Pod> pods;
    pods.reserve(pod_specs.size());
    for (auto& t : pod_specs)
      pods.push_back(Pod{std::get<0>(t), std::get<1>(t), std::get<2>(t),
                         std::get<3>(t)});
    res = solve(nodes, pods);
  }  // GIL re-acquired here; only now touch Python objects
  if (!res.feasible) return py::none();
  return py::make_tuple(res.assignments, res.score, res.consumed);
}

double score_only(size_t n_nodes, int total_gpus, double link_gbps) {
  return placement_score(n_nodes, total_gpus, link_gbps);
}

}  // namespace

// Fast deep copy for JSON-shaped objects (dict/list/scalars) using the raw CPython
// API — the store's hottest single function (every get()/update copies an object).
// Scalars are immutable in Python and shared, matching kubecore.store.json_copy.
static PyObject* fast_json_copy(PyObject* obj) {
  if (PyDict_CheckExact(obj)) {
    PyObject* out = PyDict_New();
    if (!out) return nullptr;
    PyObject *key, *value;
    Py_ssize_t pos = 0;
    while (PyDict_Next(obj, &pos, &key, &value)) {
      PyObject* copied = fast_json_copy(value);
      if (!copied || PyDict_SetItem(out, key, copied) < 0) {
        Py_XDECREF(copied);
        Py_DECREF(out);
        return nullptr;
      }
      Py_DECREF(copied);
    }
    return out;
  }
  if (PyList_CheckExact(obj)) {
    Py_ssize_t n = PyList_GET_SIZE(obj);
    PyObject* out = PyList_New(n);
    if (!out) return nullptr;
    for (Py_ssize_t i = 0; i < n; ++i) {
      PyObject* copied = fast_json_copy(PyList_GET_ITEM(obj, i));
      if (!copied) {
        Py_DECREF(out);
        return nullptr;
      }
      PyList_SET_ITEM(out, i, copied);  // steals ref
    }
    return out;
  }
  Py_INCREF(obj);
  return obj;
}

PYBIND11_MODULE(_sched, m) {
  m.def("json_copy", [](py::handle obj) {
    PyObject* out = fast_json_copy(obj.ptr());
    if (!out) throw py::error_already_set();
    return py::reinterpret_steal<py::object>(out);
  }, "fast deep copy for JSON-shaped objects");
  m.doc() = "grove_amd native gang-placement core (xGMI-aware Filter/Score)";
  m.def("place_gang", &place_gang, py::arg("nodes"), py::arg("pods"),
        "All-or-nothing gang placement; returns (assignments, score, consumed) or None");
  m.def("placement_score", &score_only, py::arg("n_nodes_used"),
        py::arg("total_gpus"), py::arg("link_gbps") = kXgmiLinkGBps);
}
