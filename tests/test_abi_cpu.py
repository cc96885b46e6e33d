"""CPU check: libveomni_hip.so loads and exports every symbol declared in
include/veomni_hip.h (no compute without a GPU)."""

import ctypes
import os
import re

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HEADER = os.path.join(REPO, "include", "veomni_hip.h")
SO = os.path.join(REPO, "veomni_amd", "libveomni_hip.so")


def header_symbols():
    src = open(HEADER).read()
    return re.findall(r"^(?:const char\*|int)\s+(vh_\w+)\s*\(", src, re.M)


@pytest.mark.skipif(not os.path.exists(SO), reason="extension not built")
def test_abi_exports_all_header_symbols():
    lib = ctypes.CDLL(SO)
    syms = header_symbols()
    assert len(syms) >= 14, syms
    for s in syms:
        assert hasattr(lib, s), f"missing export: {s}"
    lib.vh_build_info.restype = ctypes.c_char_p
    assert b"gfx950" in lib.vh_build_info()


def test_header_covers_wrappers():
    # every ctypes signature bound by hip_lib exists in the header
    from veomni_amd.ops import hip_lib  # noqa: F401

    syms = set(header_symbols())
    bound = set(re.findall(r'_sig\(lib, "(vh_\w+)"', open(os.path.join(
        REPO, "veomni_amd", "ops", "hip_lib.py")).read()))
    assert bound, "no ctypes signatures found"
    missing = bound - syms
    assert not missing, missing
