"""C-ABI surface: the library loads and exports every symbol
include/dprk.h declares (no compute without a GPU — tier rule ③)."""

import os
import re

import pytest

from dpark_amd import engine

HDR = os.path.join(os.path.dirname(__file__), "..", "include", "dprk.h")


def test_library_loads_and_exports_all_symbols():
    lib = engine.load_library()
    for sym in engine.SYMBOLS:
        assert hasattr(lib, sym), sym
    assert b"dprk" in lib.dprk_build_info()


def test_header_symbols_covered():
    """Every function declared in include/dprk.h is in engine.SYMBOLS (so
    the loads-and-exports test really covers the header)."""
    with open(HDR) as f:
        hdr = f.read()
    declared = set(re.findall(r"\b(dprk_\w+)\s*\(", hdr))
    declared.discard("dprk_ctx")
    assert declared == set(engine.SYMBOLS), (
        declared.symmetric_difference(engine.SYMBOLS))


def test_engine_fails_loudly_without_gpu_or_so(monkeypatch):
    """The product path must not fall back silently: a missing .so raises
    EngineUnavailable with a diagnostic."""
    monkeypatch.setattr(engine, "_SO", "/nonexistent/libdprk.so")
    monkeypatch.setattr(engine, "_lib", None)
    monkeypatch.setattr(engine, "_load_error", None)
    with pytest.raises(engine.EngineUnavailable):
        engine.load_library()
