import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, torch.distributed as td
td.init_process_group(backend="gloo")
rank = int(os.environ["RANK"]); world = int(os.environ["WORLD_SIZE"])
from minivite_amd import Graph, Engine, comm_id
if rank == 0:
    cid = comm_id(); t = torch.tensor(list(cid), dtype=torch.uint8)
else:
    t = torch.zeros(128, dtype=torch.uint8)
td.broadcast(t, 0)
cid = bytes(t.tolist())
g = Graph.rgg(16384, rank, world)
e = Engine(device=0, rank=rank, nranks=world, comm_id_bytes=cid)
e.load_graph(g)
e.set_trace(64)
mod, iters = e.run()
tt, tm = e.trace(iters)
if rank == 0:
    import hashlib, numpy as np, json
    print("RANK0 mod=%.17g iters=%d" % (mod, iters))
td.barrier()
if rank == 1:
    print("RANK1 mod=%.17g" % mod)
