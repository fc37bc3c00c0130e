#!/usr/bin/env python3
"""Config-5-shaped perf evidence: com-Orkut's .bin is unavailable offline
(BASELINE.md / DESIGN.md §7 note the substitution), so build a synthetic
graph with Orkut's published shape — 3,072,441 vertices, ~234 M directed
edges (avg deg ~76), power-law degrees capped at ~33k — and run the
engine on one GPU through the same -f/-b machinery (write .bin, read
back, run). Records edges/s for the irregular-degree regime the RGG
configs never reach (hub kernels + degree-sorted order + spill sizing).

Usage: python experiments/orkut_like.py [--nv N] [--avg-deg D]
"""
import argparse
import json
import os
import sys
import tempfile
import time

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)


def build_graph(nv, avg_deg, max_deg, seed=1, structure=True,
                p_intra=0.6, ncomm=100000):
    """Power-law-degree graph; with `structure`, ~p_intra of each vertex's
    edges go to members of its planted community (Zipf community sizes) —
    real clustering like a social graph, so Louvain actually iterates and
    candidate counts per vertex stay small after the first rounds."""
    rng = np.random.default_rng(seed)
    # Zipf-ish out-degrees scaled to hit avg_deg/2 before symmetrization
    raw = (1.0 / rng.power(1.6, nv)).astype(np.int64)
    raw = np.minimum(raw, max_deg // 2)
    scale = (avg_deg / 2.0) / raw.mean()
    deg = np.maximum((raw * scale).astype(np.int64), 1)
    src = np.repeat(np.arange(nv, dtype=np.int64), deg)
    if structure:
        # planted communities with Zipf sizes
        cw = 1.0 / np.arange(1, ncomm + 1) ** 0.9
        comm = rng.choice(ncomm, nv, p=cw / cw.sum())
        order_v = np.argsort(comm, kind="stable")
        cstart = np.searchsorted(comm[order_v], np.arange(ncomm))
        cend = np.searchsorted(comm[order_v], np.arange(ncomm), side="right")
        csz = cend - cstart
        intra = rng.random(src.size) < p_intra
        cs = comm[src]
        # random member of src's community (may self-hit; filtered below)
        pick = cstart[cs] + (rng.random(src.size) * csz[cs]).astype(np.int64)
        dst = np.where(intra, order_v[np.minimum(pick, len(order_v) - 1)],
                       rng.integers(0, nv, src.size, dtype=np.int64))
    else:
        dst = rng.integers(0, nv, src.size, dtype=np.int64)
    keep = src != dst
    src, dst = src[keep], dst[keep]
    u = np.concatenate([src, dst])
    v = np.concatenate([dst, src])
    order = np.lexsort((v, u))
    u, v = u[order], v[order]
    xadj = np.zeros(nv + 1, dtype=np.int64)
    np.add.at(xadj, u + 1, 1)
    xadj = np.cumsum(xadj)
    return xadj, v


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nv", type=int, default=3072441)
    ap.add_argument("--avg-deg", type=int, default=76)
    ap.add_argument("--max-deg", type=int, default=33000)
    ap.add_argument("--steps", type=int, default=3)
    ap.add_argument("--warmup", type=int, default=1)
    ap.add_argument("--random", action="store_true",
                    help="no planted communities (worst-case stress: every "
                         "neighbor a distinct candidate)")
    args = ap.parse_args()

    from minivite_amd import Graph, Engine

    t0 = time.perf_counter()
    xadj, tails = build_graph(args.nv, args.avg_deg, args.max_deg,
                              structure=not args.random)
    gen_s = time.perf_counter() - t0
    lne = len(tails)

    parts = np.array([0, args.nv], dtype=np.int64)
    g0 = Graph.from_csr(args.nv, 0, 1, parts, xadj, tails, None)
    # exercise the -f file machinery like config 5 would
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "orkutlike.bin")
        g0.write_binary(path)
        g0.free()
        g = Graph.read_binary(path, 0, 1)

    e = Engine(device=0)
    t0 = time.perf_counter()
    e.load_graph(g)
    load_s = time.perf_counter() - t0
    mod = iters = None
    for _ in range(args.warmup):
        mod, iters = e.run()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        mod, iters = e.run()
    dt = (time.perf_counter() - t0) / args.steps
    st = e.stats()
    e.destroy()
    g.free()
    print(json.dumps({
        "workload": f"orkut_like_nv{args.nv}_deg{args.avg_deg}"
                    f"_cap{args.max_deg}"
                    f"_{'random' if args.random else 'planted'}",
        "directed_edges": lne,
        "max_degree": int((xadj[1:] - xadj[:-1]).max()),
        "iterations": iters,
        "modularity": mod,
        "edge_visits_per_s": lne * iters / dt,
        "s_per_run": round(dt, 4),
        "sweep_ms_per_launch": round(st["sweep_ms"] / st["sweep_launches"],
                                     4),
        "gen_seconds": round(gen_s, 1),
        "load_seconds": round(load_s, 1),
    }), flush=True)


if __name__ == "__main__":
    main()
