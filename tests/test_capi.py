"""C API tests: compile a small C host program against
libmultiverso_amd.so and check the exact reference symbol surface
(c_api.h:16-54) end to end in a fresh process."""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

C_HOST = r"""
#include <stdio.h>
#include <stdlib.h>
#include "c_api.h"

int main(void) {
  int argc = 1; char* argv0 = (char*)"prog"; char* argv[] = {argv0};
  MV_Init(&argc, argv);
  if (MV_NumWorkers() != 1 || MV_WorkerId() != 0 || MV_ServerId() != 0)
    return 2;
  MV_Barrier();

  TableHandler at;
  MV_NewArrayTable(8, &at);
  float ones[8], out[8];
  for (int i = 0; i < 8; ++i) ones[i] = 1.0f;
  MV_AddArrayTable(at, ones, 8);
  MV_AddAsyncArrayTable(at, ones, 8);
  MV_GetArrayTable(at, out, 8);
  for (int i = 0; i < 8; ++i) if (out[i] != 2.0f) return 3;

  TableHandler mt;
  MV_NewMatrixTable(4, 3, &mt);
  float m[12], mo[12];
  for (int i = 0; i < 12; ++i) m[i] = (float)i;
  MV_AddMatrixTableAll(mt, m, 12);
  MV_GetMatrixTableAll(mt, mo, 12);
  for (int i = 0; i < 12; ++i) if (mo[i] != (float)i) return 4;

  int rows[2] = {1, 3};
  float rv[6] = {10, 10, 10, 20, 20, 20}, ro[6];
  MV_AddMatrixTableByRows(mt, rv, 6, rows, 2);
  MV_GetMatrixTableByRows(mt, ro, 6, rows, 2);
  if (ro[0] != 13.0f || ro[3] != 29.0f) return 5;
  MV_AddAsyncMatrixTableByRows(mt, rv, 6, rows, 2);

  MV_ShutDown();
  printf("CAPI_OK\n");
  return 0;
}
"""


@pytest.fixture(scope="module")
def capi_so():
    from multiverso_amd import capi
    return capi.build(verbose=True)


def test_c_host_roundtrip(capi_so, tmp_path_factory):
    tmp = tmp_path_factory.mktemp("capi")
    csrc = tmp / "host.c"
    csrc.write_text(C_HOST)
    exe = tmp / "host"
    inc = os.path.join(REPO, "multiverso_amd", "capi")
    subprocess.run(
        ["gcc", "-O0", str(csrc), f"-I{inc}", f"-L{inc}",
         "-lmultiverso_amd", "-o", str(exe)], check=True)
    env = dict(os.environ)
    env["LD_LIBRARY_PATH"] = inc + ":" + env.get("LD_LIBRARY_PATH", "")
    env["PYTHONPATH"] = REPO + ":" + env.get("PYTHONPATH", "")
    r = subprocess.run([str(exe)], env=env, capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 0, (r.returncode, r.stdout, r.stderr)
    assert "CAPI_OK" in r.stdout


def test_c_host_roundtrip_sanitized(tmp_path_factory):
    """the reference ships no sanitizer jobs; here the whole
    native host path (C host + libmultiverso_amd) runs under
    ASan+UBSan. Leaks are not checked (the embedded CPython interpreter
    frees at process exit); memory errors and UB abort the run."""
    import shutil
    import sysconfig
    import pybind11
    tmp = tmp_path_factory.mktemp("capi_asan")
    inc = os.path.join(REPO, "multiverso_amd", "capi")
    so = tmp / "libmultiverso_amd.so"
    py_inc = sysconfig.get_paths()["include"]
    libdir = sysconfig.get_config_var("LIBDIR") or "/usr/lib"
    pyver = f"python{sysconfig.get_python_version()}"
    san = ["-fsanitize=address,undefined", "-fno-sanitize-recover=all",
           "-fno-sanitize=vptr"]
    subprocess.run(
        ["g++", "-O1", "-g", "-std=c++17", "-shared", "-fPIC",
         os.path.join(inc, "c_api.cpp"), f"-I{py_inc}",
         f"-I{pybind11.get_include()}", f"-L{libdir}", f"-l{pyver}",
         "-o", str(so)] + san, check=True)
    csrc = tmp / "host.c"
    csrc.write_text(C_HOST)
    exe = tmp / "host"
    subprocess.run(
        ["gcc", "-O0", "-g", str(csrc), f"-I{inc}", f"-L{tmp}",
         "-lmultiverso_amd", "-o", str(exe)] + san, check=True)
    env = dict(os.environ)
    env["LD_LIBRARY_PATH"] = f"{tmp}:" + env.get("LD_LIBRARY_PATH", "")
    env["PYTHONPATH"] = REPO + ":" + env.get("PYTHONPATH", "")
    env["ASAN_OPTIONS"] = "detect_leaks=0:abort_on_error=1"
    r = subprocess.run([str(exe)], env=env, capture_output=True, text=True,
                       timeout=600)
    assert r.returncode == 0, (r.returncode, r.stdout[-500:], r.stderr[-2000:])
    assert "CAPI_OK" in r.stdout


CPP_HOST = r"""
#include <cstdio>
#include <vector>
#include "multiverso.hpp"

int main() {
  multiverso::SetCMDFlag("sync", "true");
  multiverso::Init();
  if (multiverso::Rank() != 0 || multiverso::Size() != 1) return 2;
  if (multiverso::NumWorkers() != 1 || multiverso::NumServers() != 1)
    return 2;
  multiverso::Barrier();

  multiverso::ArrayTableHandler arr(8);
  std::vector<float> ones(8, 1.0f), out;
  arr.Add(ones.data());
  arr.Get(out);
  for (float v : out) if (v != 1.0f) return 3;

  std::vector<float> agg(4, 2.0f);
  multiverso::Aggregate(agg);          // world 1: identity
  if (agg[0] != 2.0f) return 4;

  multiverso::MatrixTableHandler mat(6, 2);
  std::vector<float> rv = {5, 5, 7, 7}, ro(4);
  std::vector<int> rows = {1, 4};
  mat.AddByRows(rv.data(), rows);
  mat.GetByRows(ro.data(), rows);
  if (ro[0] != 5.0f || ro[2] != 7.0f) return 5;
  std::vector<float> whole(12);
  mat.GetAll(whole.data());
  if (whole[2] != 5.0f || whole[8] != 7.0f) return 6;

  multiverso::ShutDown();
  std::printf("CPPAPI_OK\n");
  return 0;
}
"""


def test_cpp_host_roundtrip(capi_so, tmp_path_factory):
    """The reference's public C++ API surface (multiverso.h:9-68) as a
    native C++ host program over multiverso.hpp."""
    tmp = tmp_path_factory.mktemp("cppapi")
    src = tmp / "host.cpp"
    src.write_text(CPP_HOST)
    exe = tmp / "host"
    inc = os.path.join(REPO, "multiverso_amd", "capi")
    subprocess.run(
        ["g++", "-O0", "-std=c++17", str(src), f"-I{inc}", f"-L{inc}",
         "-lmultiverso_amd", "-o", str(exe)], check=True)
    env = dict(os.environ)
    env["LD_LIBRARY_PATH"] = inc + ":" + env.get("LD_LIBRARY_PATH", "")
    env["PYTHONPATH"] = REPO + ":" + env.get("PYTHONPATH", "")
    r = subprocess.run([str(exe)], env=env, capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 0, (r.returncode, r.stdout, r.stderr)
    assert "CPPAPI_OK" in r.stdout
