"""The `multiverso` drop-in package (binding/python) must expose the
reference binding's API surface (binding/python/multiverso/api.py,
tables.py) backed by multiverso_amd."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "binding", "python"))

import torch


def test_shim_surface_and_roundtrip():
    import multiverso as mv
    for name in ("init", "shutdown", "barrier", "workers_num", "worker_id",
                 "server_id", "is_master_worker", "ArrayTableHandler",
                 "MatrixTableHandler"):
        assert hasattr(mv, name), name
    mv.init()
    t = mv.ArrayTableHandler(12, init_value=torch.arange(12).float())
    got = t.get()
    assert torch.allclose(torch.as_tensor(got),
                          torch.arange(12).float())
    m = mv.MatrixTableHandler(4, 3)
    m.add(torch.ones(4, 3))
    assert torch.equal(torch.as_tensor(m.get()), torch.ones(4, 3))
    mv.shutdown()


def _c_api_symbols():
    import re
    hdr = open(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "multiverso_amd", "capi",
        "c_api.h")).read()
    return set(re.findall(r"\b(MV_\w+)\s*\(", hdr))


def test_lua_binding_symbols_match_c_api():
    """The Lua FFI cdefs must only name symbols the C API exports."""
    import re
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    lua = open(os.path.join(root, "binding", "lua", "multiverso.lua")).read()
    used = set(re.findall(r"\b(MV_\w+)\s*\(", lua))
    exported = _c_api_symbols()
    assert used, "no MV_ symbols found in the Lua binding"
    assert used <= exported, used - exported


def test_csharp_binding_symbols_match_c_api():
    """Every DllImport EntryPoint must exist in c_api.h."""
    import re
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cs = open(os.path.join(root, "binding", "csharp",
                           "Multiverso.cs")).read()
    used = set(re.findall(r'EntryPoint = "(MV_\w+)"', cs))
    exported = _c_api_symbols()
    assert used, "no EntryPoints found in the C# binding"
    assert used <= exported, used - exported


def test_c_api_symbol_set_matches_reference():
    """Exact reference symbol surface (c_api.h:16-54) must be present,
    plus the documented extensions that carry the reference's C++ API
    (multiverso.h:9-68) to native hosts — nothing else."""
    reference = {
        "MV_Init", "MV_ShutDown", "MV_Barrier", "MV_NumWorkers",
        "MV_WorkerId", "MV_ServerId",
        "MV_NewArrayTable", "MV_GetArrayTable", "MV_AddArrayTable",
        "MV_AddAsyncArrayTable",
        "MV_NewMatrixTable", "MV_GetMatrixTableAll", "MV_AddMatrixTableAll",
        "MV_AddAsyncMatrixTableAll", "MV_GetMatrixTableByRows",
        "MV_AddMatrixTableByRows", "MV_AddAsyncMatrixTableByRows",
    }
    extensions = {
        "MV_Rank", "MV_Size", "MV_NumServers", "MV_Aggregate",
        "MV_SetFlag", "MV_NetBind", "MV_NetConnect",
    }
    assert _c_api_symbols() == reference | extensions
