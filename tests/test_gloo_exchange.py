"""N>1 decomposition coverage on CPU (gloo, world_size 2).

Two processes each hold one rank's slice of the RGG (built by the product
generator) and execute the engine's per-iteration exchange PROTOCOL —
ghost want-lists once, then per iteration: ghost-community alltoallv,
remote-community-info request/reply, delta routing, 2-double allreduce —
as real message passing over torch.distributed (gloo), with the same
routing rules the HIP engine uses (sorted ghost lists segmented by owner,
count matrices via allgather, reply/delta buffers aligned by request
order). The per-rank compute between exchanges runs in numpy following the
reference semantics. The run must reproduce the oracle pin bit-for-bit
(unit weights), proving the distributed protocol carries exactly the data
the algorithm needs.
"""
import json
import os
import sys

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
GOLDEN = os.path.join(os.path.dirname(__file__), "golden", "pins.json")


def _rank_main(rank, world, nv, out_q):
    import torch
    import torch.distributed as td
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", str(29601 + world))
    td.init_process_group("gloo", rank=rank, world_size=world)
    sys.path.insert(0, REPO)
    from minivite_amd import Graph

    g = Graph.rgg(nv, rank, world)
    xadj, tails, w = g.arrays()
    parts = np.array([(nv * r) // world for r in range(world + 1)], dtype=np.int64)
    base, bound = parts[rank], parts[rank + 1]
    lnv = bound - base

    def owner(c):
        return np.minimum(np.searchsorted(parts, c, side="right") - 1, world - 1)

    # ---- exchangeVertexReqs (dspl.hpp:1112-1272): sorted unique ghosts,
    # segmented by owner; role-swapped lists via alltoall ----
    remote = tails[(tails < base) | (tails >= bound)]
    ghosts = np.unique(remote)  # sorted
    recv_off = np.searchsorted(ghosts, parts)  # owner segments
    want = np.diff(recv_off)
    want_t = torch.tensor(want, dtype=torch.int64)
    matrix = [torch.zeros(world, dtype=torch.int64) for _ in range(world)]
    td.all_gather(matrix, want_t)
    send_cnt = np.array([int(matrix[r][rank]) for r in range(world)])
    send_cnt[rank] = 0

    def alltoallv(send_parts_list, recv_sizes, dtype=torch.int64):
        """send_parts_list[r] tensor for peer r; returns list of received."""
        ops = []
        recvs = [torch.zeros(int(recv_sizes[r]), dtype=dtype)
                 for r in range(world)]
        for r in range(world):
            if r == rank:
                continue
            if len(send_parts_list[r]):
                ops.append(td.P2POp(td.isend, send_parts_list[r], r))
            if recv_sizes[r]:
                ops.append(td.P2POp(td.irecv, recvs[r], r))
        if ops:
            for op in td.batch_isend_irecv(ops):
                op.wait()
        return recvs

    ghost_segs = [torch.from_numpy(ghosts[recv_off[r]:recv_off[r + 1]].copy())
                  for r in range(world)]
    ghost_segs[rank] = torch.zeros(0, dtype=torch.int64)
    svdata_segs = alltoallv(ghost_segs, send_cnt)
    svdata = np.concatenate([t.numpy() for t in svdata_segs])

    # translated tails: local index or lnv + ghost slot
    is_local = (tails >= base) & (tails < bound)
    tidx = np.where(is_local, tails - base, lnv + np.searchsorted(ghosts, tails))

    # ---- distInitLouvain ----
    deg = np.diff(xadj)
    vdeg = np.zeros(lnv)
    np.add.at(vdeg, np.repeat(np.arange(lnv), deg), w)
    cinfo_size = np.ones(lnv, dtype=np.int64)
    cinfo_deg = vdeg.copy()
    local_w = torch.tensor([vdeg.sum()])
    td.all_reduce(local_w)
    constant = 1.0 / float(local_w.item())
    curr = np.arange(base, bound, dtype=np.int64)

    prev_mod, iters = -1.0, 0
    mods = []
    while True:
        iters += 1
        # halo #1a: export comms of svdata, receive ghost comms
        sc = curr[svdata - base]
        sc_segs, pos = [], 0
        for r in range(world):
            n = send_cnt[r]
            sc_segs.append(torch.from_numpy(sc[pos:pos + n].copy()))
            pos += n
        gc_segs = alltoallv(sc_segs, want)
        ghost_comm = np.concatenate([t.numpy() for t in gc_segs]) \
            if ghosts.size else np.zeros(0, dtype=np.int64)

        # candidate remote communities
        cand = np.concatenate([ghost_comm, curr])
        cand = np.unique(cand[(cand < base) | (cand >= bound)])
        rc_bounds = np.searchsorted(cand, parts)
        req_cnt_mine = np.diff(rc_bounds)
        rt = torch.tensor(req_cnt_mine, dtype=torch.int64)
        m2 = [torch.zeros(world, dtype=torch.int64) for _ in range(world)]
        td.all_gather(m2, rt)
        incoming = np.array([int(m2[r][rank]) for r in range(world)])
        incoming[rank] = 0
        req_segs = alltoallv(
            [torch.from_numpy(cand[rc_bounds[r]:rc_bounds[r + 1]].copy())
             if r != rank else torch.zeros(0, dtype=torch.int64)
             for r in range(world)], incoming)
        req_ids = np.concatenate([t.numpy() for t in req_segs])
        # reply with (size, degree)
        reply_sz = cinfo_size[req_ids - base] if req_ids.size else np.zeros(0, dtype=np.int64)
        reply_dg = cinfo_deg[req_ids - base] if req_ids.size else np.zeros(0)
        pos = 0
        rs_segs, rd_segs = [], []
        for r in range(world):
            n = incoming[r]
            rs_segs.append(torch.from_numpy(reply_sz[pos:pos + n].copy()))
            rd_segs.append(torch.from_numpy(reply_dg[pos:pos + n].copy()))
            pos += n
        rc_size = np.concatenate([t.numpy() for t in
                                  alltoallv(rs_segs, req_cnt_mine)])
        rc_deg = np.concatenate([t.numpy() for t in
                                 alltoallv(rd_segs, req_cnt_mine,
                                           torch.float64)])

        # ---- the sweep (numpy restatement of dspl.hpp:276-405) ----
        comm_ext = np.concatenate([curr, ghost_comm])
        target = np.empty(lnv, dtype=np.int64)
        cw = np.zeros(lnv)
        cupd_sz = np.zeros(lnv, dtype=np.int64)
        cupd_dg = np.zeros(lnv)
        rcu_sz = np.zeros(cand.size, dtype=np.int64)
        rcu_dg = np.zeros(cand.size)

        def cinfo_of(c):
            if base <= c < bound:
                return cinfo_size[c - base], cinfo_deg[c - base]
            q = np.searchsorted(cand, c)
            return rc_size[q], rc_deg[q]

        for i in range(lnv):
            cc = curr[i]
            ccs, ccd = cinfo_of(cc)
            e0, e1 = xadj[i], xadj[i + 1]
            if e0 == e1:
                target[i] = cc
                continue
            keys, acc = [], []
            c0 = selfloop = 0.0
            for e in range(e0, e1):
                t = tidx[e]
                if t == i:
                    selfloop += w[e]
                tc = comm_ext[t]
                if tc == cc:
                    c0 += w[e]
                    continue
                if keys and keys[-1] == tc:
                    acc[-1] += w[e]
                else:
                    try:
                        k = keys.index(tc)
                        acc[k] += w[e]
                    except ValueError:
                        keys.append(tc)
                        acc.append(w[e])
            cw[i] += c0
            eix = c0 - selfloop
            ax = ccd - vdeg[i]
            best_g, best_y, best_sz = 0.0, cc, ccs
            for y, eiy in zip(keys, acc):
                ysz, ydg = cinfo_of(y)
                gain = 2.0 * (eiy - eix) - 2.0 * vdeg[i] * (ydg - ax) * constant
                if gain > best_g or (gain == best_g and gain != 0.0
                                     and y < best_y):
                    best_g, best_y, best_sz = gain, y, ysz
            if best_sz == 1 and ccs == 1 and best_y > cc:
                best_y = cc
            target[i] = best_y
            if best_y != cc:
                if base <= cc < bound:
                    cupd_sz[cc - base] -= 1
                    cupd_dg[cc - base] -= vdeg[i]
                else:
                    q = np.searchsorted(cand, cc)
                    rcu_sz[q] -= 1
                    rcu_dg[q] -= vdeg[i]
                if base <= best_y < bound:
                    cupd_sz[best_y - base] += 1
                    cupd_dg[best_y - base] += vdeg[i]
                else:
                    q = np.searchsorted(cand, best_y)
                    rcu_sz[q] += 1
                    rcu_dg[q] += vdeg[i]

        cinfo_size += cupd_sz
        cinfo_deg += cupd_dg

        # halo #2: route deltas to owners (aligned with the request order)
        ds_segs, dd_segs = [], []
        for r in range(world):
            s, t2 = rc_bounds[r], rc_bounds[r + 1]
            ds_segs.append(torch.from_numpy(rcu_sz[s:t2].copy()))
            dd_segs.append(torch.from_numpy(rcu_dg[s:t2].copy()))
        got_sz = np.concatenate([t.numpy() for t in
                                 alltoallv(ds_segs, incoming)])
        got_dg = np.concatenate([t.numpy() for t in
                                 alltoallv(dd_segs, incoming,
                                           torch.float64)])
        if req_ids.size:
            np.add.at(cinfo_size, req_ids - base, got_sz)
            np.add.at(cinfo_deg, req_ids - base, got_dg)

        # modularity
        part = torch.tensor([cw.sum(), (cinfo_deg * cinfo_deg).sum()])
        td.all_reduce(part)
        mod = abs(float(part[0]) * constant -
                  float(part[1]) * constant * constant)
        mods.append(mod)
        if mod - prev_mod < 1e-6:
            break
        prev_mod = max(mod, -1.0)
        curr = target.copy()
        if iters > 64:
            break

    out_q.put((rank, iters, prev_mod, [float(m).hex() for m in mods]))
    td.destroy_process_group()


@pytest.mark.parametrize("nv,world", [(16384, 2), (16384, 4), (32768, 8)])
def test_gloo_protocol_matches_pin(nv, world):
    import torch.multiprocessing as mp
    pins = json.load(open(GOLDEN))
    pin = pins[f"rgg_n{nv}_p{world}_unit"]
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_rank_main, args=(r, world, nv, q))
             for r in range(world)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(world):
        r = q.get(timeout=600)
        res[r[0]] = r[1:]
    for p in procs:
        p.join(timeout=60)
    iters, mod, mods = res[0]
    assert iters == pin["iters"]
    assert float(mod).hex() == pin["final_mod_hex"]
    assert mods == pin["iter_mod_hex"]
    assert res[0][2] == res[1][2]  # both ranks agree on every modularity
