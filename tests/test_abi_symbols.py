"""The C-ABI library must export every function include/wukong_abi.h
declares (DESIGN.md section 1 boundary; no compute calls — loadable on a
box with no GPU)."""
import ctypes
import os
import re

HDR = os.path.join(os.path.dirname(__file__), "..", "include", "wukong_abi.h")
SO = os.path.join(os.path.dirname(__file__), "..", "wukong_amd",
                  "libwukong_hip.so")


def test_every_declared_symbol_exported():
    src = open(HDR).read()
    src = re.sub(r"/\*.*?\*/", "", src, flags=re.S)   # strip comments
    names = re.findall(r"\b(wk_[a-z0-9_]+)\s*\(", src)
    # typedefs like wk_engine_t are not functions; the regex above only
    # matches identifiers directly followed by '(' (declarations).
    assert len(set(names)) > 25, names
    lib = ctypes.CDLL(SO)
    missing = [n for n in set(names) if not hasattr(lib, n)]
    assert not missing, missing
