"""ctypes mirror of the C-ABI (include/minivite_hip.h).

Graph construction runs on the host and works anywhere; Engine needs an
MI355X (mv_engine_create fails loudly without one — no CPU fallback).
"""
import ctypes
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB = None

COMM_ID_BYTES = 128


class MvStats(ctypes.Structure):
    _fields_ = [
        ("total_ms", ctypes.c_double),
        ("sweep_ms", ctypes.c_double),
        ("sweep_launches", ctypes.c_int64),
        ("halo_ms", ctypes.c_double),
        ("setup_ms", ctypes.c_double),
        ("edges_local", ctypes.c_int64),
        ("iters", ctypes.c_int),
    ]


def lib():
    global _LIB
    if _LIB is None:
        path = os.path.join(_DIR, "libminivite.so")
        if not os.path.exists(path):
            raise RuntimeError(
                "libminivite.so not built — run `make -C minivite_amd/csrc` "
                "or __graft_entry__.build()")
        L = ctypes.CDLL(path)
        i64, f64, vp = ctypes.c_int64, ctypes.c_double, ctypes.c_void_p
        pi64 = ctypes.POINTER(i64)
        pf64 = ctypes.POINTER(f64)
        L.mv_graph_rgg.restype = vp
        L.mv_graph_rgg.argtypes = [i64, ctypes.c_int, ctypes.c_int,
                                   ctypes.c_int, f64, ctypes.c_uint64]
        L.mv_graph_read_binary.restype = vp
        L.mv_graph_read_binary.argtypes = [ctypes.c_char_p, ctypes.c_int,
                                           ctypes.c_int, ctypes.c_int]
        L.mv_graph_from_csr.restype = vp
        L.mv_graph_from_csr.argtypes = [i64, ctypes.c_int, ctypes.c_int, pi64,
                                        i64, i64, pi64, pi64, pf64]
        L.mv_graph_write_binary.restype = ctypes.c_int
        L.mv_graph_write_binary.argtypes = [vp, ctypes.c_char_p]
        L.mv_graph_free.restype = None
        L.mv_graph_free.argtypes = [vp]
        for name in ("mv_graph_nv", "mv_graph_lnv", "mv_graph_lne"):
            getattr(L, name).restype = i64
            getattr(L, name).argtypes = [vp]
        for name in ("mv_graph_parts", "mv_graph_xadj", "mv_graph_tails"):
            getattr(L, name).restype = pi64
            getattr(L, name).argtypes = [vp]
        L.mv_graph_weights.restype = pf64
        L.mv_graph_weights.argtypes = [vp]
        L.mv_comm_id.restype = ctypes.c_int
        L.mv_comm_id.argtypes = [ctypes.c_void_p]
        L.mv_lb_create.restype = vp
        L.mv_lb_create.argtypes = [ctypes.c_int]
        L.mv_lb_destroy.restype = None
        L.mv_lb_destroy.argtypes = [vp]
        L.mv_engine_create_lb.restype = vp
        L.mv_engine_create_lb.argtypes = [ctypes.c_int, ctypes.c_int,
                                          ctypes.c_int, vp]
        L.mv_engine_create.restype = vp
        L.mv_engine_create.argtypes = [ctypes.c_int, ctypes.c_int, ctypes.c_int,
                                       ctypes.c_void_p]
        L.mv_engine_destroy.restype = None
        L.mv_engine_destroy.argtypes = [vp]
        L.mv_engine_load_graph.restype = ctypes.c_int
        L.mv_engine_load_graph.argtypes = [vp, vp]
        L.mv_engine_run.restype = f64
        L.mv_engine_run.argtypes = [vp, f64, f64, ctypes.POINTER(ctypes.c_int)]
        L.mv_engine_set_trace.restype = None
        L.mv_engine_set_trace.argtypes = [vp, pi64, pf64, ctypes.c_int]
        L.mv_engine_get_stats.restype = None
        L.mv_engine_get_stats.argtypes = [vp, ctypes.POINTER(MvStats)]
        _LIB = L
    return _LIB


def _i64p(a):
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))


def _f64p(a):
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_double))


def comm_id():
    """RCCL unique id bytes (call on rank 0, broadcast out of band)."""
    buf = ctypes.create_string_buffer(COMM_ID_BYTES)
    if lib().mv_comm_id(buf) != 0:
        raise RuntimeError("mv_comm_id failed")
    return bytes(buf.raw)


class Graph:
    """This rank's slice of the 1-D partitioned CSR (host memory)."""

    def __init__(self, handle, owns=True):
        if not handle:
            raise RuntimeError("graph construction failed")
        self.h = handle
        self._owns = owns

    @classmethod
    def rgg(cls, nv, rank=0, nranks=1, unit_weight=True,
            random_edge_percent=0.0, random_edge_seed=7177):
        return cls(lib().mv_graph_rgg(nv, rank, nranks,
                                      1 if unit_weight else 0,
                                      random_edge_percent, random_edge_seed))

    @classmethod
    def read_binary(cls, path, rank=0, nranks=1, balanced=False):
        return cls(lib().mv_graph_read_binary(path.encode(), rank, nranks,
                                              1 if balanced else 0))

    @classmethod
    def from_csr(cls, nv, rank, nranks, parts, xadj, tails, weights=None):
        parts = np.ascontiguousarray(parts, dtype=np.int64)
        xadj = np.ascontiguousarray(xadj, dtype=np.int64)
        tails = np.ascontiguousarray(tails, dtype=np.int64)
        wp = None
        if weights is not None:
            weights = np.ascontiguousarray(weights, dtype=np.float64)
            wp = _f64p(weights)
        return cls(lib().mv_graph_from_csr(nv, rank, nranks, _i64p(parts),
                                           len(xadj) - 1, len(tails),
                                           _i64p(xadj), _i64p(tails), wp))

    @property
    def nv(self):
        return lib().mv_graph_nv(self.h)

    @property
    def lnv(self):
        return lib().mv_graph_lnv(self.h)

    @property
    def lne(self):
        return lib().mv_graph_lne(self.h)

    def arrays(self):
        lnv, lne = self.lnv, self.lne
        xadj = np.ctypeslib.as_array(lib().mv_graph_xadj(self.h), (lnv + 1,)).copy()
        tails = np.ctypeslib.as_array(lib().mv_graph_tails(self.h), (lne,)).copy()
        w = np.ctypeslib.as_array(lib().mv_graph_weights(self.h), (lne,)).copy()
        return xadj, tails, w

    def write_binary(self, path):
        if lib().mv_graph_write_binary(self.h, path.encode()) != 0:
            raise RuntimeError("mv_graph_write_binary failed")

    def free(self):
        if getattr(self, "h", None) and self._owns:
            lib().mv_graph_free(self.h)
            self.h = None

    def __del__(self):
        try:
            self.free()
        except Exception:
            pass


class LoopbackSession:
    """Hardware-validation harness: nranks engines in ONE process on ONE
    device (see minivite_hip.h). Test-only; product multi-GPU is RCCL."""

    def __init__(self, nranks):
        self.nranks = nranks
        self.h = lib().mv_lb_create(nranks)
        if not self.h:
            raise RuntimeError("mv_lb_create failed")

    def destroy(self):
        if getattr(self, "h", None):
            lib().mv_lb_destroy(self.h)
            self.h = None

    def __del__(self):
        try:
            self.destroy()
        except Exception:
            pass


class Engine:
    """The GPU Louvain engine (one per process, one GPU)."""

    def __init__(self, device=0, rank=0, nranks=1, comm_id_bytes=None):
        cid = None
        if comm_id_bytes is not None:
            cid = ctypes.create_string_buffer(comm_id_bytes, COMM_ID_BYTES)
        self.h = lib().mv_engine_create(device, rank, nranks, cid)
        if not self.h:
            raise RuntimeError(
                "mv_engine_create failed — an MI355X GPU is required; "
                "there is no CPU fallback")
        self._trace_buf = None

    @classmethod
    def loopback(cls, session, rank, device=0):
        """Rank `rank` of a LoopbackSession (all ranks share `device`)."""
        obj = cls.__new__(cls)
        obj.h = lib().mv_engine_create_lb(device, rank, session.nranks,
                                          session.h)
        if not obj.h:
            raise RuntimeError("mv_engine_create_lb failed")
        obj._trace_buf = None
        return obj

    def load_graph(self, g):
        if lib().mv_engine_load_graph(self.h, g.h) != 0:
            raise RuntimeError("mv_engine_load_graph failed")
        self._lnv = g.lnv

    def set_trace(self, cap=256):
        self._trace_buf = np.zeros(cap * self._lnv, dtype=np.int64)
        self._trace_mod = np.zeros(cap, dtype=np.float64)
        self._trace_cap = cap
        lib().mv_engine_set_trace(self.h, _i64p(self._trace_buf),
                                  _f64p(self._trace_mod), cap)

    def run(self, lower=-1.0, thresh=1e-6):
        it = ctypes.c_int(0)
        mod = lib().mv_engine_run(self.h, lower, thresh, ctypes.byref(it))
        return mod, it.value

    def trace(self, iters):
        n = min(iters, self._trace_cap)
        return (self._trace_buf.reshape(self._trace_cap, self._lnv)[:n],
                self._trace_mod[:n])

    def stats(self):
        s = MvStats()
        lib().mv_engine_get_stats(self.h, ctypes.byref(s))
        return {f: getattr(s, f) for f, _ in s._fields_}

    def destroy(self):
        if getattr(self, "h", None):
            lib().mv_engine_destroy(self.h)
            self.h = None

    def __del__(self):
        try:
            self.destroy()
        except Exception:
            pass
