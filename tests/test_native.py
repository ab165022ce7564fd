"""Native C++ placement core cross-check vs the Python reference implementation."""
import pytest

from grove_amd.scheduler.placement import NodeFree, PodRequest, place_gang

try:
    from grove_amd.scheduler import _sched
except ImportError:
    _sched = None

needs_native = pytest.mark.skipif(_sched is None, reason="_sched.so not built")


def mk_nodes(spec):
    return [NodeFree(n, cpu, mem, list(range(g)), pods)
            for (n, cpu, mem, g, pods) in spec]


def native_place(nodes, pods):
    flat_nodes = [(n.name, n.cpu_milli, float(n.mem_bytes), list(n.gpu_ids), n.pods,
                   float(n.link_gbps)) for n in nodes]
    flat_pods = [(p.name, p.cpu_milli, float(p.mem_bytes), p.gpus) for p in pods]
    return _sched.place_gang(flat_nodes, flat_pods)


CASES = [
    # one empty 8-GPU node, 8x1-GPU gang -> single node, score 153
    ([("a", 100000, 1e12, 8, 100)], [(f"p{i}", 1000, 1e9, 1) for i in range(8)]),
    # two nodes, one partially full -> pick the emptier that fits whole gang
    ([("a", 100000, 1e12, 4, 100), ("b", 100000, 1e12, 8, 100)],
     [(f"p{i}", 1000, 1e9, 1) for i in range(8)]),
    # doesn't fit on one -> spread over 2
    ([("a", 100000, 1e12, 6, 100), ("b", 100000, 1e12, 6, 100)],
     [(f"p{i}", 1000, 1e9, 1) for i in range(8)]),
    # impossible
    ([("a", 100000, 1e12, 2, 100)], [(f"p{i}", 1000, 1e9, 1) for i in range(8)]),
    # cpu-bound fit
    ([("a", 3000, 1e12, 0, 100), ("b", 100000, 1e12, 0, 100)],
     [(f"p{i}", 2000, 1e9, 0) for i in range(4)]),
]


@pytest.mark.parametrize("case", range(len(CASES)))
def test_python_placement(case):
    node_spec, pod_spec = CASES[case]
    nodes = mk_nodes(node_spec)
    pods = [PodRequest(n, c, m, g) for (n, c, m, g) in pod_spec]
    res = place_gang(nodes, pods)
    _check(res, node_spec, pod_spec)


@needs_native
@pytest.mark.parametrize("case", range(len(CASES)))
def test_native_matches_python(case):
    node_spec, pod_spec = CASES[case]
    py_nodes = mk_nodes(node_spec)
    py_pods = [PodRequest(n, c, m, g) for (n, c, m, g) in pod_spec]
    py_res = place_gang([n.clone() for n in py_nodes], py_pods)
    nat = native_place(py_nodes, py_pods)
    if py_res is None:
        assert nat is None
        return
    assert nat is not None
    _, py_score = py_res
    nat_assignments, nat_score, consumed = nat
    assert nat_score == pytest.approx(py_score)
    # same node-spread cardinality
    py_nodes_used = {a.node for a in py_res[0]}
    nat_nodes_used = {a[1] for a in nat_assignments}
    assert len(nat_nodes_used) == len(py_nodes_used)
    _check((py_res[0], py_score), node_spec, pod_spec)
    # native assignment validity: every pod placed once, gpu ids disjoint per node
    assert sorted(a[0] for a in nat_assignments) == sorted(p[0] for p in pod_spec)
    per_node: dict = {}
    for (_pod, node, gpus) in nat_assignments:
        taken = per_node.setdefault(node, set())
        for g in gpus:
            assert g not in taken
            taken.add(g)


def _check(res, node_spec, pod_spec):
    cap = {n: g for (n, _c, _m, g, _p) in node_spec}
    total_gpus_needed = sum(g for (_n, _c, _m, g) in pod_spec)
    total_avail = sum(cap.values())
    if total_gpus_needed > total_avail:
        assert res is None
        return
    if res is None:
        return  # may be infeasible for other resources
    assignments, score = res
    assert len(assignments) == len(pod_spec)
    used: dict = {}
    for a in assignments:
        used.setdefault(a.node, []).extend(a.gpu_ids)
    for node, gpus in used.items():
        assert len(gpus) == len(set(gpus)) <= cap[node]
    if len(used) == 1 and total_gpus_needed > 1:
        assert score == pytest.approx(153.0)
    elif total_gpus_needed > 1:
        assert score < 153.0


@needs_native
def test_native_consumption_applied():
    nodes = mk_nodes([("a", 100000, 1e12, 8, 100)])
    pods = [PodRequest("p0", 1000, 1e9, 4)]
    _assignments, _score, consumed = native_place(nodes, pods)
    (name, cpu, _mem, gpu_ids, pods_left, _link) = consumed[0]
    assert name == "a" and cpu == 99000 and len(gpu_ids) == 4 and pods_left == 99


@needs_native
def test_native_measured_link_bandwidth_score():
    """Score reports the MEASURED min per-link bandwidth of the chosen pool, and
    ties between equally-packed pools prefer the faster fabric (both solvers)."""
    from grove_amd.scheduler.placement import place_gang
    nodes = [NodeFree("slow", 100000, 1e12, list(range(8)), 100, link_gbps=100.0),
             NodeFree("fast", 100000, 1e12, list(range(8)), 100, link_gbps=150.0)]
    pods = [PodRequest(f"p{i}", 1000, 1e9, 1) for i in range(8)]
    res_py = place_gang([n.clone() for n in nodes], pods)
    assert res_py is not None
    assert res_py[1] == 150.0
    assert all(a.node == "fast" for a in res_py[0])
    res_nat = native_place([n.clone() for n in nodes], pods)
    assert res_nat is not None
    assert res_nat[1] == 150.0
    # single-GPU gang: fabric max = links x measured per-link bw
    one = place_gang([NodeFree("m", 1000, 1e9, [0], 10, link_gbps=120.0)],
                     [PodRequest("p", 100, 1e6, 1)])
    assert one[1] == 120.0 * 7


def test_rect_tile_map_host_model():
    """CPU model of tile_map_rect (gpuwork.hip): for every supported grid the
    mapping must be a BIJECTION onto the tile grid, and the blocks resident on
    one XCD (bid % 8) within a 256-block window must cover exactly an
    m_per_xcd x n_win rectangle (the L2-footprint contract the +15-18% at 8192^3
    rests on). Pure math — no GPU needed."""
    def rect(bid, tiles_m, tiles_n, m_per_xcd, n_win):
        mwin_tiles = 8 * m_per_xcd
        win, idx = bid >> 8, bid & 255
        xcd, slot = idx & 7, idx >> 3
        nwins = tiles_n // n_win
        tm = (win // nwins) * mwin_tiles + xcd * m_per_xcd + slot % m_per_xcd
        tn = (win % nwins) * n_win + slot // m_per_xcd
        return tm, tn

    for tiles_m, tiles_n in [(16, 16), (32, 32), (32, 64), (64, 32), (64, 64)]:
        mpx = min(4, max(1, tiles_m // 8))
        nw = 256 // (8 * mpx)
        if tiles_m % (8 * mpx) or tiles_n % nw:
            continue  # host launch falls back to the column map
        seen = set()
        per_xcd = {}
        for bid in range(tiles_m * tiles_n):
            tm, tn = rect(bid, tiles_m, tiles_n, mpx, nw)
            assert 0 <= tm < tiles_m and 0 <= tn < tiles_n, (tiles_m, tiles_n, bid)
            assert (tm, tn) not in seen, f"duplicate tile at bid {bid}"
            seen.add((tm, tn))
            per_xcd.setdefault((bid >> 8, bid & 7), set()).add((tm, tn))
        assert len(seen) == tiles_m * tiles_n  # bijection
        for (win, xcd), tiles in per_xcd.items():
            ms = {t[0] for t in tiles}
            ns = {t[1] for t in tiles}
            assert len(ms) == mpx and len(ns) == nw and \
                len(tiles) == mpx * nw, "per-XCD footprint must be a rectangle"
