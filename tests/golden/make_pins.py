#!/usr/bin/env python3
"""Generate tests/golden/pins.json from the oracle, and (when the reference
tree + oracle/_ref harness exist, i.e. in the dev container) validate every
pin bit-for-bit against the REAL reference first.

The pins anchor parity on any machine (the GPU box has no /root/reference):
they hold, per config, the directed edge count, per-rank graph array hashes,
iteration count, per-iteration modularity (hex floats, exact) and
per-iteration targetComm sha256 prefixes, all produced by the oracle and —
for every unit-weight config — verified identical to the reference's own
dump harness output (see repo history for the validation run).

Usage: python tests/golden/make_pins.py [--validate]
"""
import json
import os
import struct
import subprocess
import sys
import tempfile

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, REPO)
from oracle.oracle import OracleGraph, louvain, sha  # noqa: E402

CONFIGS = [
    # (nv, p, unit_weight)
    (16384, 1, True),
    (16384, 2, True),
    (16384, 4, True),
    (16384, 8, True),
    (16384, 1, False),
    (16384, 2, False),
    (65536, 1, True),
    (65536, 4, True),
    (32768, 8, True),
    (16384, 16, True),
]

REF_HARNESS = os.path.join(REPO, "oracle", "_ref", "harness_dump")


def ref_dump(nv, p, unit):
    d = tempfile.mkdtemp(prefix="refdump_")
    env = dict(os.environ)
    env["LD_PRELOAD"] = "/usr/lib/x86_64-linux-gnu/libstdc++.so.6"
    env["PATH"] = "/opt/conda/bin:" + env.get("PATH", "")
    cmd = ["mpiexec", "-n", str(p), REF_HARNESS, str(nv), d] + ([] if unit else ["w"])
    subprocess.run(cmd, check=True, env=env, capture_output=True)
    return d


def load_graph_dump(path):
    with open(path, "rb") as f:
        lnv, lne = struct.unpack("<qq", f.read(16))
        xadj = np.fromfile(f, dtype=np.int64, count=lnv + 1)
        el = np.fromfile(f, dtype=np.dtype([("tail", "<i8"), ("w", "<f8")]), count=lne)
    return xadj, el["tail"].copy(), el["w"].copy()


def main():
    validate = "--validate" in sys.argv and os.path.exists(REF_HARNESS)
    pins = {}
    for nv, p, unit in CONFIGS:
        key = f"rgg_n{nv}_p{p}_{'unit' if unit else 'w'}"
        g = OracleGraph.rgg(nv, p, unit_weight=unit)
        mod, iters, tt, tm = louvain(g, trace=True)
        parts = [(nv * r) // p for r in range(p + 1)]
        entry = {
            "nv": nv, "p": p, "unit": unit,
            "ne": int(g.total_edges()),
            "iters": int(iters),
            "final_mod_hex": float(mod).hex(),
            "iter_mod_hex": [float(m).hex() for m in tm],
            "iter_target_sha": [sha(tt[k]) for k in range(iters)],
            "graph_sha": [
                [sha(a) for a in g.rank_arrays(r)] for r in range(p)
            ],
        }
        if validate:
            d = ref_dump(nv, p, unit)
            for r in range(p):
                xa, ta, wa = load_graph_dump(os.path.join(d, f"graph_r{r}.bin"))
                oxa, ota, owa = g.rank_arrays(r)
                assert np.array_equal(xa, oxa) and np.array_equal(ta, ota) \
                    and np.array_equal(wa, owa), f"{key} rank {r} graph mismatch"
            ref_mods = [float.fromhex(l.split()[3])
                        for l in open(os.path.join(d, "trace_r0.txt"))
                        if l.startswith("iter")]
            assert len(ref_mods) == iters, f"{key} iters {iters} != ref {len(ref_mods)}"
            for k in range(1, iters + 1):
                full = np.concatenate([
                    np.fromfile(os.path.join(d, f"target_i{k}_r{r}.bin"), dtype=np.int64)
                    for r in range(p)])
                assert np.array_equal(full, tt[k - 1]), f"{key} iter {k} targets differ"
            if unit:
                assert all(tm[k] == ref_mods[k] for k in range(iters)), \
                    f"{key} unit-weight modularity bits differ"
            else:
                assert all(abs(tm[k] - ref_mods[k]) < 1e-9 for k in range(iters))
            entry["validated_against_reference"] = True
            print(f"{key}: VALIDATED vs reference ({iters} iters, ne={entry['ne']})")
        else:
            print(f"{key}: oracle-only ({iters} iters, ne={entry['ne']})")
        pins[key] = entry
        g.free()

    out = os.path.join(REPO, "tests", "golden", "pins.json")
    with open(out, "w") as f:
        json.dump(pins, f, indent=1)
    print("wrote", out)


if __name__ == "__main__":
    main()
