"""The C-ABI library loads (no GPU needed to dlopen) and exports every
symbol include/darray_hip.h declares; opcode tables agree across the
header, the Python mirror and the oracle."""
import ctypes
import os
import re

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HDR = os.path.join(ROOT, "include", "darray_hip.h")
SO = os.path.join(ROOT, "distributedarrays_jl_amd", "libdarray_hip.so")


@pytest.fixture(scope="module")
def built():
    if not os.path.exists(SO):
        import __graft_entry__
        __graft_entry__.build()
    return ctypes.CDLL(SO)


def header_symbols():
    txt = open(HDR).read()
    # function declarations: "int da_xxx(...)" / "const char* da_xxx(" /
    # "uint64_t da_xxx("
    return sorted(set(re.findall(
        r"^(?:int|const char\*|uint64_t)\s+(da_\w+)\s*\(", txt, re.M)))


def test_all_header_symbols_exported(built):
    syms = header_symbols()
    assert len(syms) >= 30
    missing = [s for s in syms if not hasattr(built, s)]
    assert not missing, "not exported: %r" % missing


def test_ffi_covers_header():
    from distributedarrays_jl_amd import _ffi
    syms = set(header_symbols())
    bound = set(_ffi._sigs)
    assert syms == bound, ("header/_ffi mismatch: only-header=%r only-ffi=%r"
                           % (syms - bound, bound - syms))


def test_opcode_tables_agree_with_header():
    from distributedarrays_jl_amd import _opcodes
    txt = open(HDR).read()
    m = re.search(r"enum da_mapop \{(.*?)\};", txt, re.S)
    names = re.findall(r"DA_OP_(\w+)", m.group(1))
    names = [n for n in names if n != "_N"]
    assert [n.lower() for n in names] == _opcodes.MAP_OPS
    m = re.search(r"enum da_map2op \{(.*?)\};", txt, re.S)
    names = [n for n in re.findall(r"DA_OP2_(\w+)", m.group(1)) if n != "_N"]
    assert [n.lower() for n in names] == _opcodes.MAP2_OPS


def test_opcode_tables_agree_with_oracle():
    from distributedarrays_jl_amd import _opcodes
    from oracle import ops as oops
    for name in _opcodes.MAP_OPS:
        assert name in oops.MAP_OPS, name
    for name in _opcodes.MAP2_OPS:
        assert name in oops.MAP2_OPS, name
    assert set(_opcodes.RED_OPS) == set(oops.RED_OPS)
    assert set(_opcodes.RED_FS) == set(oops.MAPRED_FS)


def test_errstr_without_init(built):
    # calling a compute entry before da_init must fail loudly, not crash
    rc = built.da_synchronize()
    assert rc < 0
    built.da_errstr.restype = ctypes.c_char_p
    assert b"da_init" in built.da_errstr(rc)


def test_integration_stubs_exist(built):
    """Every ccall symbol named in INTEGRATION.md's Julia stub block must
    be exported by the library."""
    txt = open(os.path.join(ROOT, "INTEGRATION.md")).read()
    syms = set(re.findall(r":(da_\w+)", txt))
    assert len(syms) >= 15
    missing = [s for s in syms if not hasattr(built, s)]
    assert not missing, missing
