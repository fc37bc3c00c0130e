"""C-ABI surface checks that need no GPU: the shared library loads and
exports every symbol include/minivite_hip.h declares, and the engine
constructor fails LOUDLY (no silent CPU fallback) when no GPU exists."""
import ctypes
import os
import re

import pytest

import minivite_amd
from minivite_amd import Graph

HEADER = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                      "include", "minivite_hip.h")


def header_symbols():
    txt = open(HEADER).read()
    # function declarations: name directly before '('
    names = re.findall(r"\b(mv_[a-z_0-9]+)\s*\(", txt)
    return sorted(set(names))


def test_library_exports_header_symbols():
    L = minivite_amd.lib()
    missing = []
    for sym in header_symbols():
        try:
            getattr(L, sym)
        except AttributeError:
            missing.append(sym)
    assert not missing, f"libminivite.so missing symbols: {missing}"


def test_engine_fails_loudly_without_gpu():
    try:
        import torch
        if torch.cuda.is_available():
            pytest.skip("GPU present")
    except ImportError:
        pass
    from minivite_amd import Engine
    with pytest.raises(RuntimeError, match="no CPU fallback"):
        Engine(device=0)


def test_graph_accessors():
    g = Graph.rgg(16384, 0, 1)
    try:
        assert g.nv == 16384
        assert g.lnv == 16384
        xadj, tails, w = g.arrays()
        assert xadj[-1] == g.lne == len(tails) == len(w)
        assert (tails >= 0).all() and (tails < 16384).all()
        assert (w == 1.0).all()
    finally:
        g.free()
