"""Pre-validate the driver's 8-GPU SCALE shape on ONE GPU via loopback:
RGG n=2^24 partitioned 8 ways (BASELINE configs[2]), all 8 engines in one
process. Checks completion, cross-rank agreement and run-to-run
determinism — not perf (loopback serializes the exchanges)."""
import os
import sys
import threading

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from minivite_amd import Graph, Engine, LoopbackSession  # noqa: E402


def run_once(nv, world, graphs):
    ses = LoopbackSession(world)
    out = {}
    errs = []

    def rank_main(r):
        try:
            e = Engine.loopback(ses, r, device=0)
            e.load_graph(graphs[r])
            mod, iters = e.run()
            st = e.stats()
            out[r] = (mod, iters, st["halo_ms"], st["setup_ms"])
            e.destroy()
        except Exception as ex:  # pragma: no cover
            errs.append((r, repr(ex)))

    threads = [threading.Thread(target=rank_main, args=(r,))
               for r in range(world)]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=1200)
    ses.destroy()
    assert not errs, errs
    mods = {r: v[0] for r, v in out.items()}
    iters = {v[1] for v in out.values()}
    assert len(iters) == 1, iters
    assert len({float(m).hex() for m in mods.values()}) == 1, mods
    return mods[0], iters.pop()


def main():
    nv, world = 1 << 24, 8
    graphs = []
    for r in range(world):
        graphs.append(Graph.rgg(nv, r, world))
    m1, i1 = run_once(nv, world, graphs)
    m2, i2 = run_once(nv, world, graphs)
    assert (float(m1).hex(), i1) == (float(m2).hex(), i2), (m1, i1, m2, i2)
    for g in graphs:
        g.free()
    print(f"scale-shape OK: n=2^24 p=8 loopback, mod={m1:.6f} iters={i1}, "
          f"deterministic across reruns")


if __name__ == "__main__":
    main()
