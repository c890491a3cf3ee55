"""Execute the Lua and C# binding declarations against the real
library.

The image has no LuaJIT and no mono/dotnet, so the bindings themselves
cannot run (COVERAGE.md §2.7).  What CAN run is the exact marshalling
contract each binding declares: this module parses the FFI cdef block
out of binding/lua/multiverso.lua and the DllImport signatures out of
binding/csharp/Multiverso.cs, converts each declared prototype to a
ctypes prototype (same arity, same type classes LuaJIT-FFI / .NET
P/Invoke would use), and drives the reference C-host roundtrip
(test_capi.C_HOST, itself mirroring reference binding/lua/test.lua:16-71)
through those prototypes against libmultiverso_amd.so.  A cdef typo, a
wrong argument order, or a type-size mismatch that the symbol-drift
check (test_binding_shim.py) cannot see fails here.
"""

import ctypes
import os
import re

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

# ---------------------------------------------------------------- parsers

_C_TYPES = {
    "void": None,
    "int": ctypes.c_int,
    "int*": ctypes.POINTER(ctypes.c_int),
    "float*": ctypes.POINTER(ctypes.c_float),
    "char*": ctypes.c_char_p,
    "char**": ctypes.POINTER(ctypes.c_char_p),
    "TableHandler": ctypes.c_void_p,
    "TableHandler*": ctypes.POINTER(ctypes.c_void_p),
}


def _norm_c_type(t):
    t = t.replace("const", " ").strip()
    t = re.sub(r"\s*\*\s*", "*", t)
    t = re.sub(r"\s+", " ", t).strip()
    return t


def _parse_c_param(p):
    """'int* argc' / 'char* argv[]' / 'int row_ids[]' -> ctypes type."""
    p = p.strip()
    extra_ptr = 0
    if p.endswith("[]"):
        p = p[:-2].rstrip()
        extra_ptr = 1
    m = re.match(r"^(.*?)\s*\*?\s*([A-Za-z_]\w*)$", p)
    assert m, p
    t = _norm_c_type(p[: p.rfind(m.group(2))])
    t += "*" * extra_ptr
    assert t in _C_TYPES, f"unmapped C type {t!r} in param {p!r}"
    return _C_TYPES[t]


def parse_lua_cdefs():
    """FFI cdef block -> {symbol: (restype, [argtypes])}."""
    src = open(os.path.join(REPO, "binding", "lua", "multiverso.lua")).read()
    m = re.search(r"ffi\.cdef\[\[(.*?)\]\]", src, re.S)
    assert m, "no ffi.cdef block"
    sigs = {}
    body = re.sub(r"typedef[^;]*;", "", m.group(1))
    for decl in re.finditer(r"([\w\*\s]+?)\b(MV_\w+)\s*\(([^)]*)\)\s*;",
                            body, re.S):
        ret, name, params = decl.groups()
        ret = _norm_c_type(ret)
        assert ret in _C_TYPES, f"unmapped return {ret!r} for {name}"
        params = params.strip()
        args = ([] if params in ("", "void")
                else [_parse_c_param(p) for p in params.split(",")])
        sigs[name] = (_C_TYPES[ret], args)
    return sigs


_CS_TYPES = {
    "void": None,
    "int": ctypes.c_int,
    "ref int": ctypes.POINTER(ctypes.c_int),
    "string": ctypes.c_char_p,
    "string[]": ctypes.POINTER(ctypes.c_char_p),
    "float[]": ctypes.POINTER(ctypes.c_float),
    "int[]": ctypes.POINTER(ctypes.c_int),
    "IntPtr": ctypes.c_void_p,
    "out IntPtr": ctypes.POINTER(ctypes.c_void_p),
}


def parse_csharp_imports():
    """DllImport declarations -> {entry point: (restype, [argtypes])}."""
    src = open(os.path.join(REPO, "binding", "csharp",
                            "Multiverso.cs")).read()
    sigs = {}
    pat = re.compile(
        r'\[DllImport\(Lib,\s*EntryPoint\s*=\s*"(MV_\w+)"\)\]\s*'
        r"(?:public|private|internal)\s+static\s+extern\s+"
        r"([\w\[\]]+)\s+\w+\s*\(([^)]*)\)\s*;", re.S)
    for ep, ret, params in pat.findall(src):
        assert ret in _CS_TYPES, f"unmapped C# return {ret!r} for {ep}"
        args = []
        for p in [q.strip() for q in params.split(",") if q.strip()]:
            # 'ref int argc' / 'float[] data' / 'out IntPtr handler'
            m = re.match(r"^((?:ref |out )?[\w\[\]]+)\s+\w+$", p)
            assert m, p
            t = m.group(1)
            assert t in _CS_TYPES, f"unmapped C# type {t!r} in {ep}"
            args.append(_CS_TYPES[t])
        sigs[ep] = (_CS_TYPES[ret], args)
    return sigs


# ------------------------------------------------------------- roundtrip


@pytest.fixture(scope="module")
def capi_lib():
    from multiverso_amd import capi
    so = capi.build()
    return ctypes.CDLL(str(so), mode=ctypes.RTLD_GLOBAL)


def _bind(lib, sigs):
    fns = {}
    for name, (restype, argtypes) in sigs.items():
        f = getattr(lib, name)  # raises if the symbol is missing
        f.restype = restype
        f.argtypes = argtypes
        fns[name] = f
    return fns


def _roundtrip(fn):
    """The C-host scenario, driven through the binding's declared
    prototypes (reference binding/lua/test.lua:16-71 semantics)."""
    argc = ctypes.c_int(0)
    fn["MV_Init"](ctypes.byref(argc), None)
    assert fn["MV_NumWorkers"]() == 1
    assert fn["MV_WorkerId"]() == 0
    assert fn["MV_ServerId"]() == 0
    assert fn["MV_Rank"]() == 0
    assert fn["MV_Size"]() == 1
    assert fn["MV_NumServers"]() == 1
    fn["MV_SetFlag"](b"log_level", b"info")
    fn["MV_Barrier"]()

    at = ctypes.c_void_p()
    fn["MV_NewArrayTable"](8, ctypes.byref(at))
    ones = (ctypes.c_float * 8)(*([1.0] * 8))
    fn["MV_AddArrayTable"](at, ones, 8)
    fn["MV_AddAsyncArrayTable"](at, ones, 8)
    out = (ctypes.c_float * 8)()
    fn["MV_GetArrayTable"](at, out, 8)
    assert list(out) == [2.0] * 8

    mt = ctypes.c_void_p()
    fn["MV_NewMatrixTable"](4, 3, ctypes.byref(mt))
    m = (ctypes.c_float * 12)(*[float(i) for i in range(12)])
    fn["MV_AddMatrixTableAll"](mt, m, 12)
    fn["MV_AddAsyncMatrixTableAll"](mt, m, 12)
    mo = (ctypes.c_float * 12)()
    fn["MV_GetMatrixTableAll"](mt, mo, 12)
    assert list(mo) == [2.0 * i for i in range(12)]

    rows = (ctypes.c_int * 2)(1, 3)
    rv = (ctypes.c_float * 6)(10, 10, 10, 20, 20, 20)
    fn["MV_AddMatrixTableByRows"](mt, rv, 6, rows, 2)
    ro = (ctypes.c_float * 6)()
    fn["MV_GetMatrixTableByRows"](mt, ro, 6, rows, 2)
    # row 1 was [6,8,10] doubled; +10 each.  row 3 was [18,20,22]; +20.
    assert list(ro) == [16.0, 18.0, 20.0, 38.0, 40.0, 42.0]
    fn["MV_AddAsyncMatrixTableByRows"](mt, rv, 6, rows, 2)
    fn["MV_GetMatrixTableByRows"](mt, ro, 6, rows, 2)
    assert list(ro) == [26.0, 28.0, 30.0, 58.0, 60.0, 62.0]

    agg = (ctypes.c_float * 4)(1, 2, 3, 4)
    fn["MV_Aggregate"](agg, 4)  # world 1: identity
    assert list(agg) == [1.0, 2.0, 3.0, 4.0]

    fn["MV_ShutDown"]()


def test_lua_declared_prototypes_execute(capi_lib):
    sigs = parse_lua_cdefs()
    assert len(sigs) >= 17  # the reference surface at minimum
    _roundtrip(_bind(capi_lib, sigs))


def test_csharp_declared_prototypes_execute(capi_lib):
    sigs = parse_csharp_imports()
    assert len(sigs) >= 17
    _roundtrip(_bind(capi_lib, sigs))


def test_lua_and_csharp_declare_identical_shapes():
    """Shared symbols must agree on arity and on every positional type
    class between the two bindings (both must match the C ABI, so any
    disagreement is a bug in at least one)."""
    lua, cs = parse_lua_cdefs(), parse_csharp_imports()
    shared = set(lua) & set(cs)
    assert len(shared) >= 17
    for name in sorted(shared):
        assert lua[name] == cs[name], name
