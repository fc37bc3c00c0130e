"""ctypes wrapper for liboracle.so (the CPU restatement of the reference).

TEST INFRASTRUCTURE ONLY: importable from tests/, __graft_entry__.smoke()
and bench.py's cpu_baseline leg. The product package (minivite_amd) must
never import this module.
"""
import ctypes
import hashlib
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
_LIB = None


def lib():
    global _LIB
    if _LIB is None:
        path = os.path.join(_DIR, "liboracle.so")
        if not os.path.exists(path):
            raise RuntimeError(
                "liboracle.so not built — run `make -C oracle` or __graft_entry__.build()")
        L = ctypes.CDLL(path)
        L.orc_reseeder.restype = ctypes.c_uint32
        L.orc_reseeder.argtypes = [ctypes.c_uint32]
        L.orc_lcg_fill.restype = None
        L.orc_lcg_fill.argtypes = [ctypes.c_uint32, ctypes.c_int64, ctypes.c_int,
                                   ctypes.POINTER(ctypes.c_double)]
        L.orc_graph_new.restype = ctypes.c_void_p
        L.orc_graph_new.argtypes = [ctypes.c_int64, ctypes.c_int]
        L.orc_graph_set_parts.restype = None
        L.orc_graph_set_parts.argtypes = [ctypes.c_void_p, ctypes.POINTER(ctypes.c_int64)]
        L.orc_graph_set_rank_csr.restype = None
        L.orc_graph_set_rank_csr.argtypes = [
            ctypes.c_void_p, ctypes.c_int, ctypes.c_int64, ctypes.c_int64,
            ctypes.POINTER(ctypes.c_int64), ctypes.POINTER(ctypes.c_int64),
            ctypes.POINTER(ctypes.c_double)]
        L.orc_rgg_generate.restype = ctypes.c_void_p
        L.orc_rgg_generate.argtypes = [ctypes.c_int64, ctypes.c_int, ctypes.c_int]
        L.orc_graph_lne.restype = ctypes.c_int64
        L.orc_graph_lne.argtypes = [ctypes.c_void_p, ctypes.c_int]
        L.orc_graph_lnv.restype = ctypes.c_int64
        L.orc_graph_lnv.argtypes = [ctypes.c_void_p, ctypes.c_int]
        L.orc_graph_xadj.restype = ctypes.POINTER(ctypes.c_int64)
        L.orc_graph_xadj.argtypes = [ctypes.c_void_p, ctypes.c_int]
        L.orc_graph_tails.restype = ctypes.POINTER(ctypes.c_int64)
        L.orc_graph_tails.argtypes = [ctypes.c_void_p, ctypes.c_int]
        L.orc_graph_weights.restype = ctypes.POINTER(ctypes.c_double)
        L.orc_graph_weights.argtypes = [ctypes.c_void_p, ctypes.c_int]
        L.orc_graph_free.restype = None
        L.orc_graph_free.argtypes = [ctypes.c_void_p]
        L.orc_louvain.restype = ctypes.c_double
        L.orc_louvain.argtypes = [ctypes.c_void_p, ctypes.c_double, ctypes.c_int,
                                  ctypes.POINTER(ctypes.c_int),
                                  ctypes.POINTER(ctypes.c_int64),
                                  ctypes.POINTER(ctypes.c_double)]
        _LIB = L
    return _LIB


def _i64p(a):
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_int64))


def _f64p(a):
    return a.ctypes.data_as(ctypes.POINTER(ctypes.c_double))


class OracleGraph:
    """P-rank partitioned CSR, held by the oracle."""

    def __init__(self, handle, nv, p):
        self.h = handle
        self.nv = nv
        self.p = p

    @classmethod
    def rgg(cls, nv, p, unit_weight=True):
        h = lib().orc_rgg_generate(nv, p, 1 if unit_weight else 0)
        return cls(h, nv, p)

    @classmethod
    def from_csr(cls, nv, p, parts, rank_csrs):
        """rank_csrs: list of (xadj, tails, weights-or-None) numpy arrays."""
        h = lib().orc_graph_new(nv, p)
        parts = np.ascontiguousarray(parts, dtype=np.int64)
        lib().orc_graph_set_parts(h, _i64p(parts))
        for r, (xadj, tails, w) in enumerate(rank_csrs):
            xadj = np.ascontiguousarray(xadj, dtype=np.int64)
            tails = np.ascontiguousarray(tails, dtype=np.int64)
            wp = None
            if w is not None:
                w = np.ascontiguousarray(w, dtype=np.float64)
                wp = _f64p(w)
            lib().orc_graph_set_rank_csr(h, r, len(xadj) - 1, len(tails),
                                         _i64p(xadj), _i64p(tails), wp)
        return cls(h, nv, p)

    def lne(self, r):
        return lib().orc_graph_lne(self.h, r)

    def lnv(self, r):
        return lib().orc_graph_lnv(self.h, r)

    def total_edges(self):
        return sum(self.lne(r) for r in range(self.p))

    def rank_arrays(self, r):
        lnv, lne = self.lnv(r), self.lne(r)
        xadj = np.ctypeslib.as_array(lib().orc_graph_xadj(self.h, r), (lnv + 1,)).copy()
        tails = np.ctypeslib.as_array(lib().orc_graph_tails(self.h, r), (lne,)).copy()
        w = np.ctypeslib.as_array(lib().orc_graph_weights(self.h, r), (lne,)).copy()
        return xadj, tails, w

    def free(self):
        if self.h:
            lib().orc_graph_free(self.h)
            self.h = None

    def __del__(self):
        try:
            self.free()
        except Exception:
            pass


def louvain(g, thresh=1e-6, max_iters=0, trace=False, trace_cap=256):
    """Run the oracle Louvain. Returns (mod, iters[, target_trace, mod_trace])."""
    it = ctypes.c_int(0)
    if trace:
        tt = np.zeros(trace_cap * g.nv, dtype=np.int64)
        tm = np.zeros(trace_cap, dtype=np.float64)
        mod = lib().orc_louvain(g.h, thresh, max_iters or trace_cap, ctypes.byref(it),
                                _i64p(tt), _f64p(tm))
        n = it.value
        return mod, n, tt.reshape(trace_cap, g.nv)[:n], tm[:n]
    mod = lib().orc_louvain(g.h, thresh, max_iters, ctypes.byref(it), None, None)
    return mod, it.value


def sha(a):
    return hashlib.sha256(np.ascontiguousarray(a).tobytes()).hexdigest()[:16]
