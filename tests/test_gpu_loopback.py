"""Multi-rank HIP path on ONE GPU via the loopback transport.

RCCL refuses two ranks on one device, so the p>1 engine code — the
view-mode K4 sweeps (engine.hip k4_sweep_mr / k4_sweep_iter1_mr / the hub
kernels), k8 community gathers, view builds, candidate filter +
sort/unique, K9 replies, delta compaction, per-sender delta application,
and the whole deep-pipelined per-iteration exchange protocol
(dspl.hpp:497-1103 equivalents) — is executed here with the collectives
replaced by device-to-device copies + host barriers (mv_lb_session).
Every kernel and every byte layout is the production p>1 path; only
ncclSend/Recv itself is substituted. Parity: bit-exact vs the
reference-pinned oracle fixtures (tests/golden/pins.json) or the oracle
on identical partitioned inputs.
"""
import json
import os
import threading

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

GOLDEN = os.path.join(os.path.dirname(__file__), "golden", "pins.json")


def _run_loopback(nv, world, unit=True, trace_cap=64):
    from minivite_amd import Graph, Engine, LoopbackSession
    ses = LoopbackSession(world)
    results = {}
    errors = []

    def rank_main(r):
        try:
            g = Graph.rgg(nv, r, world, unit_weight=unit)
            e = Engine.loopback(ses, r, device=0)
            e.load_graph(g)
            e.set_trace(trace_cap)
            mod, iters = e.run()
            tt, tm = e.trace(iters)
            results[r] = (mod, iters, tt.copy(),
                          [float(m).hex() for m in tm])
            e.destroy()
            g.free()
        except Exception as ex:  # pragma: no cover
            errors.append((r, repr(ex)))

    threads = [threading.Thread(target=rank_main, args=(r,))
               for r in range(world)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=600)
    ses.destroy()
    assert not errors, errors
    assert len(results) == world
    return results


def _check_vs_pin(results, pin, world, exact_mod=True):
    from oracle.oracle import sha
    nv = pin["nv"]
    iters = results[0][1]
    assert iters == pin["iters"]
    for r in range(world):  # every rank agrees on the global result
        assert results[r][1] == iters
        if exact_mod:
            assert float(results[r][0]).hex() == pin["final_mod_hex"], \
                f"rank {r}"
            assert results[r][3] == pin["iter_mod_hex"]
        else:
            assert abs(results[r][0]
                       - float.fromhex(pin["final_mod_hex"])) < 1e-9
    parts = [(nv * r) // world for r in range(world + 1)]
    for k in range(min(iters, 64)):
        full = np.zeros(nv, dtype=np.int64)
        for r in range(world):
            full[parts[r]:parts[r + 1]] = results[r][2][k]
        assert sha(full) == pin["iter_target_sha"][k], f"iteration {k+1}"


@pytest.mark.parametrize("world", [2, 4, 8])
def test_loopback_parity_unit(world):
    pins = json.load(open(GOLDEN))
    pin = pins[f"rgg_n16384_p{world}_unit"]
    results = _run_loopback(16384, world, unit=True)
    _check_vs_pin(results, pin, world, exact_mod=True)


def test_loopback_parity_p8_n32768():
    """The round-end 8-GPU topology's pin at a larger graph."""
    pins = json.load(open(GOLDEN))
    pin = pins["rgg_n32768_p8_unit"]
    results = _run_loopback(32768, 8, unit=True)
    _check_vs_pin(results, pin, 8, exact_mod=True)


def test_loopback_parity_weighted_p2():
    """-w path at p=2: communities bit-exact, modularity < 1e-9 (the
    reference's own cross-thread accumulation granularity)."""
    pins = json.load(open(GOLDEN))
    pin = pins["rgg_n16384_p2_w"]
    results = _run_loopback(16384, 2, unit=False)
    _check_vs_pin(results, pin, 2, exact_mod=False)


def test_loopback_parity_no_overlap_p4():
    """The non-overlapped halo path (MV_NO_OVERLAP) must produce the same
    bits — overlap is pure scheduling."""
    os.environ["MV_NO_OVERLAP"] = "1"
    try:
        pins = json.load(open(GOLDEN))
        pin = pins["rgg_n16384_p4_unit"]
        results = _run_loopback(16384, 4, unit=True)
        _check_vs_pin(results, pin, 4, exact_mod=True)
    finally:
        del os.environ["MV_NO_OVERLAP"]


def test_loopback_parity_no_delta_p4():
    """Full-resend #1a (MV_NO_DELTA, the reference's wire behavior) must
    produce the same bits as the delta-compacted default."""
    os.environ["MV_NO_DELTA"] = "1"
    try:
        pins = json.load(open(GOLDEN))
        pin = pins["rgg_n16384_p4_unit"]
        results = _run_loopback(16384, 4, unit=True)
        _check_vs_pin(results, pin, 4, exact_mod=True)
    finally:
        del os.environ["MV_NO_DELTA"]


def _zipf_graph(nv, max_deg, seed=7, weighted=False):
    rng = np.random.default_rng(seed)
    deg = np.minimum((2.0 / rng.power(2.0, nv)).astype(np.int64), max_deg)
    src = np.repeat(np.arange(nv), deg)
    dst = rng.integers(0, nv, src.size)
    keep = src != dst
    src, dst = src[keep], dst[keep]
    u = np.concatenate([src, dst])
    v = np.concatenate([dst, src])
    if weighted:
        w0 = rng.uniform(0.01, 1.0, src.size)
        w = np.concatenate([w0, w0])
    else:
        w = None
    order = np.lexsort((v, u))
    u, v = u[order], v[order]
    if w is not None:
        w = w[order]
    xadj = np.zeros(nv + 1, dtype=np.int64)
    np.add.at(xadj, u + 1, 1)
    xadj = np.cumsum(xadj)
    return xadj, v, w


@pytest.mark.parametrize("weighted", [False, True])
def test_loopback_skewed_p2(weighted):
    """Skewed (hub-heavy) graph partitioned 2 ways: exercises the
    multi-rank wave-per-vertex hub kernel (unit) / the spill path (-w)
    under the view encoding; engine vs oracle on identical CSRs."""
    from minivite_amd import Graph, Engine, LoopbackSession
    from oracle.oracle import OracleGraph, louvain, sha
    nv, world = 50000, 2
    xadj, tails, w = _zipf_graph(nv, 5000, weighted=weighted)
    parts = np.array([(nv * r) // world for r in range(world + 1)],
                     dtype=np.int64)
    csrs = []
    for r in range(world):
        lo, hi = parts[r], parts[r + 1]
        xa = (xadj[lo:hi + 1] - xadj[lo]).copy()
        ta = tails[xadj[lo]:xadj[hi]].copy()
        wa = None if w is None else w[xadj[lo]:xadj[hi]].copy()
        csrs.append((xa, ta, wa))
    og = OracleGraph.from_csr(nv, world, parts, csrs)
    omod, oiters, ott, otm = louvain(og, trace=True, trace_cap=300)
    og.free()

    ses = LoopbackSession(world)
    results = {}
    errors = []

    def rank_main(r):
        try:
            xa, ta, wa = csrs[r]
            g = Graph.from_csr(nv, r, world, parts, xa, ta, wa)
            e = Engine.loopback(ses, r, device=0)
            e.load_graph(g)
            e.set_trace(300)
            mod, iters = e.run()
            tt, _ = e.trace(iters)
            results[r] = (mod, iters, tt.copy())
            e.destroy()
            g.free()
        except Exception as ex:  # pragma: no cover
            errors.append((r, repr(ex)))

    import threading
    threads = [threading.Thread(target=rank_main, args=(r,))
               for r in range(world)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=600)
    ses.destroy()
    assert not errors, errors
    iters = results[0][1]
    assert iters == oiters
    if weighted:
        assert abs(results[0][0] - omod) < 1e-9
    else:
        assert float(results[0][0]).hex() == float(omod).hex()
    for k in range(min(iters, 300)):
        full = np.zeros(nv, dtype=np.int64)
        for r in range(world):
            full[parts[r]:parts[r + 1]] = results[r][2][k]
        assert sha(full) == sha(ott[k]), f"iteration {k+1}"


def test_loopback_balanced_read_p4():
    """-b edge-balanced reader (find_balanced_num_edges partition,
    graph.hpp:416-461) feeding the ENGINE at p=4 on one GPU: engine vs
    oracle on the identical balanced partition of the same .bin."""
    import tempfile
    from minivite_amd import Graph, Engine, LoopbackSession, lib
    from oracle.oracle import OracleGraph, louvain, sha
    nv, world = 16384, 4
    g0 = Graph.rgg(nv, 0, 1)
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "g.bin")
        g0.write_binary(path)
        g0.free()
        graphs, parts = [], None
        for r in range(world):
            gr = Graph.read_binary(path, r, world, balanced=True)
            if parts is None:
                pp = lib().mv_graph_parts(gr.h)
                parts = np.array([pp[k] for k in range(world + 1)],
                                 dtype=np.int64)
            graphs.append(gr)
        og = OracleGraph.from_csr(nv, world, parts,
                                  [g.arrays() for g in graphs])
        omod, oiters, ott, otm = louvain(og, trace=True)
        og.free()

        ses = LoopbackSession(world)
        results = {}
        errors = []

        def rank_main(r):
            try:
                e = Engine.loopback(ses, r, device=0)
                e.load_graph(graphs[r])
                e.set_trace(64)
                mod, iters = e.run()
                tt, _ = e.trace(iters)
                results[r] = (mod, iters, tt.copy())
                e.destroy()
            except Exception as ex:  # pragma: no cover
                errors.append((r, repr(ex)))

        threads = [threading.Thread(target=rank_main, args=(r,))
                   for r in range(world)]
        for t in threads:
            t.start()
        for t in threads:
            t.join(timeout=600)
        ses.destroy()
        for g in graphs:
            g.free()
    assert not errors, errors
    iters = results[0][1]
    assert iters == oiters
    assert float(results[0][0]).hex() == float(omod).hex()
    for k in range(min(iters, 64)):
        full = np.zeros(nv, dtype=np.int64)
        for r in range(world):
            full[parts[r]:parts[r + 1]] = results[r][2][k]
        assert sha(full) == sha(ott[k]), f"iteration {k+1}"


def _loopback_vs_oracle(nv, world, csrs, parts, trace_cap=64,
                        exact_mod=True):
    """Run `world` loopback engines on the given per-rank CSRs and compare
    per-iteration targets + modularity against the oracle on the same
    partition."""
    from minivite_amd import Graph, Engine, LoopbackSession
    from oracle.oracle import OracleGraph, louvain, sha
    og = OracleGraph.from_csr(nv, world, parts, csrs)
    omod, oiters, ott, otm = louvain(og, trace=True, trace_cap=trace_cap)
    og.free()
    ses = LoopbackSession(world)
    results = {}
    errors = []

    def rank_main(r):
        try:
            xa, ta, wa = csrs[r]
            g = Graph.from_csr(nv, r, world, parts, xa, ta, wa)
            e = Engine.loopback(ses, r, device=0)
            e.load_graph(g)
            e.set_trace(trace_cap)
            mod, iters = e.run()
            tt, _ = e.trace(iters)
            results[r] = (mod, iters, tt.copy())
            e.destroy()
            g.free()
        except Exception as ex:  # pragma: no cover
            errors.append((r, repr(ex)))

    threads = [threading.Thread(target=rank_main, args=(r,))
               for r in range(world)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=900)
    ses.destroy()
    assert not errors, errors
    iters = results[0][1]
    assert iters == oiters
    if exact_mod:
        assert float(results[0][0]).hex() == float(omod).hex()
    else:
        assert abs(results[0][0] - omod) < 1e-9
    for k in range(min(iters, trace_cap)):
        full = np.zeros(nv, dtype=np.int64)
        for r in range(world):
            full[parts[r]:parts[r + 1]] = results[r][2][k]
        assert sha(full) == sha(ott[k]), f"iteration {k+1}"


def test_loopback_large_p4_n1048576():
    """n=2^20 RGG at p=4 through the engines: larger ghost sets, multi-
    chunk SELL, grid-stride sweeps and realloc-growth paths under the
    pipelined halo (the pins only cover n<=32768)."""
    from minivite_amd import Graph
    nv, world = 1 << 20, 4
    parts = np.array([(nv * r) // world for r in range(world + 1)],
                     dtype=np.int64)
    csrs = []
    for r in range(world):
        g = Graph.rgg(nv, r, world)
        csrs.append(g.arrays())
        g.free()
    _loopback_vs_oracle(nv, world, csrs, parts)


def test_loopback_random_edges_p4():
    """configs[3]-shaped input: RGG + 4% random long-range edges at p=4 —
    stresses alltoallv volume and the delta-compaction full/compact mix."""
    from minivite_amd import Graph
    nv, world = 65536, 4
    parts = np.array([(nv * r) // world for r in range(world + 1)],
                     dtype=np.int64)
    csrs = []
    for r in range(world):
        g = Graph.rgg(nv, r, world, random_edge_percent=4.0)
        csrs.append(g.arrays())
        g.free()
    _loopback_vs_oracle(nv, world, csrs, parts)


def test_loopback_p3_nonpow2():
    """Non-power-of-two rank count (the reference accepts any nprocs for
    -f inputs): p=3 engines on a random flat graph."""
    rng = np.random.default_rng(23)
    nv, world = 30000, 3
    m = nv * 5
    u = rng.integers(0, nv, m)
    v = rng.integers(0, nv, m)
    uu = np.concatenate([u, v])
    vv = np.concatenate([v, u])
    order = np.lexsort((vv, uu))
    uu, vv = uu[order], vv[order]
    xadj = np.zeros(nv + 1, dtype=np.int64)
    np.add.at(xadj, uu + 1, 1)
    xadj = np.cumsum(xadj)
    parts = np.array([0, 9000, 21000, 30000], dtype=np.int64)
    csrs = []
    for r in range(world):
        lo, hi = parts[r], parts[r + 1]
        xa = (xadj[lo:hi + 1] - xadj[lo]).copy()
        ta = vv[xadj[lo]:xadj[hi]].copy()
        csrs.append((xa, ta, None))
    _loopback_vs_oracle(nv, world, csrs, parts, trace_cap=256)


@pytest.mark.parametrize("trial", [0, 1, 2, 3, 4])
def test_loopback_fuzz(trial):
    """Randomized small graphs (self-loops, parallel edges, isolated
    vertices, mixed weights) x rank counts {2,3,4,6,8}: the multi-rank
    engine must match the oracle bit-for-bit on arbitrary partitioned
    inputs."""
    rng = np.random.default_rng(1000 + trial)
    nv = int(rng.integers(2000, 8000))
    world = [2, 3, 4, 6, 8][trial]
    unit = trial not in (1, 3)
    m = nv * int(rng.integers(3, 9))
    u = rng.integers(0, nv, m)
    v = rng.integers(0, nv, m)
    loops = rng.integers(0, nv, max(nv // 30, 1))
    uu = np.concatenate([u, v, loops, u[:m // 5]])
    vv = np.concatenate([v, u, loops, v[:m // 5]])
    if unit:
        w = None
    else:
        w = rng.uniform(0.01, 1.0, uu.size)
    order = np.lexsort((vv, uu))
    uu, vv = uu[order], vv[order]
    if w is not None:
        w = w[order]
    xadj = np.zeros(nv + 1, dtype=np.int64)
    np.add.at(xadj, uu + 1, 1)
    xadj = np.cumsum(xadj)
    cuts = np.sort(rng.choice(np.arange(1, nv), world - 1, replace=False))
    parts = np.concatenate([[0], cuts, [nv]]).astype(np.int64)
    csrs = []
    for r in range(world):
        lo, hi = parts[r], parts[r + 1]
        xa = (xadj[lo:hi + 1] - xadj[lo]).copy()
        ta = vv[xadj[lo]:xadj[hi]].copy()
        wa = None if w is None else w[xadj[lo]:xadj[hi]].copy()
        csrs.append((xa, ta, wa))
    _loopback_vs_oracle(nv, world, csrs, parts, trace_cap=256,
                        exact_mod=unit)


def test_loopback_deterministic_p4():
    """Two identical p=4 loopback runs agree bit-for-bit (per-sender
    in-order delta application — run-to-run determinism at nranks > 2)."""
    r1 = _run_loopback(16384, 4, unit=True)
    r2 = _run_loopback(16384, 4, unit=True)
    for r in range(4):
        assert float(r1[r][0]).hex() == float(r2[r][0]).hex()
        assert r1[r][1] == r2[r][1]
        assert np.array_equal(r1[r][2], r2[r][2])
