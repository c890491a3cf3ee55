// multiverso_amd C API implementation.
//
// Reference surface: include/multiverso/c_api.h:16-54 /
// src/c_api.cpp:10-92. The reference delegated to its C++ Zoo; this
// implementation embeds the Python runtime (pybind11 embed) so a C/Lua/C#
// host drives the identical MI355X-native collective data plane. All
// float-only, matching the reference.
//
// Threading: every entry point holds the GIL for its duration (the
// reference C API is likewise a blocking, caller-thread API).

#include <pybind11/embed.h>
#include <pybind11/numpy.h>

#include <cstring>
#include <memory>
#include <vector>

#include "c_api.h"

namespace py = pybind11;

namespace {

// Deliberately leaked: destroying the embedded interpreter at process
// exit races with torch/HIP static teardown (observed SIGSEGV at exit).
py::scoped_interpreter* g_interp = nullptr;

struct Handle {
  py::object table;  // ArrayTable or MatrixTable
};

py::module_ mv() { return py::module_::import("multiverso_amd"); }

// Some entry points are legal BEFORE MV_Init (SetFlag, NetBind/Connect
// — reference multiverso.cpp:48-68); they must boot the embedded
// interpreter themselves.
void ensure_interp() {
  if (!Py_IsInitialized()) g_interp = new py::scoped_interpreter();
}

py::array_t<float> wrap(float* data, int size) {
  // zero-copy view of caller memory
  return py::array_t<float>({(py::ssize_t)size}, {sizeof(float)}, data,
                            py::none());
}

py::object to_tensor(float* data, int size) {
  auto torch = py::module_::import("torch");
  return torch.attr("from_numpy")(wrap(data, size));
}

}  // namespace

extern "C" {

void MV_Init(int* argc, char* argv[]) {
  ensure_interp();
  py::gil_scoped_acquire gil;
  py::list args;
  if (argc && argv)
    for (int i = 0; i < *argc; ++i) args.append(std::string(argv[i]));
  mv().attr("init")(py::arg("args") = args);
}

void MV_ShutDown() {
  {
    py::gil_scoped_acquire gil;
    mv().attr("shutdown")();
  }
  // keep the interpreter alive: tables may still hold references and the
  // reference semantics allow re-Init within a process.
}

void MV_Barrier() {
  py::gil_scoped_acquire gil;
  mv().attr("barrier")();
}

int MV_NumWorkers() {
  py::gil_scoped_acquire gil;
  return mv().attr("workers_num")().cast<int>();
}

int MV_WorkerId() {
  py::gil_scoped_acquire gil;
  return mv().attr("worker_id")().cast<int>();
}

int MV_ServerId() {
  py::gil_scoped_acquire gil;
  return mv().attr("server_id")().cast<int>();
}

// ---- Array table ----

void MV_NewArrayTable(int size, TableHandler* out) {
  py::gil_scoped_acquire gil;
  auto* h = new Handle{mv().attr("ArrayTable")(size)};
  *out = h;
}

void MV_GetArrayTable(TableHandler handler, float* data, int size) {
  py::gil_scoped_acquire gil;
  auto& t = static_cast<Handle*>(handler)->table;
  py::object got = t.attr("get")();
  got = got.attr("cpu")().attr("contiguous")();
  py::array_t<float> arr = got.attr("numpy")().cast<py::array_t<float>>();
  std::memcpy(data, arr.data(), sizeof(float) * size);
}

void MV_AddArrayTable(TableHandler handler, float* data, int size) {
  py::gil_scoped_acquire gil;
  auto& t = static_cast<Handle*>(handler)->table;
  t.attr("add")(to_tensor(data, size)).attr("wait")();
}

void MV_AddAsyncArrayTable(TableHandler handler, float* data, int size) {
  py::gil_scoped_acquire gil;
  auto& t = static_cast<Handle*>(handler)->table;
  // payload is copied to the device before the async collective returns,
  // so the caller may reuse `data` immediately (reference AddAsync
  // contract).
  t.attr("add")(to_tensor(data, size).attr("clone")(),
                py::arg("async_op") = true);
}

// ---- Matrix table ----

void MV_NewMatrixTable(int num_row, int num_col, TableHandler* out) {
  py::gil_scoped_acquire gil;
  auto* h = new Handle{mv().attr("MatrixTable")(num_row, num_col)};
  *out = h;
}

void MV_GetMatrixTableAll(TableHandler handler, float* data, int size) {
  py::gil_scoped_acquire gil;
  auto& t = static_cast<Handle*>(handler)->table;
  py::object got = t.attr("get")().attr("reshape")(-1);
  got = got.attr("cpu")().attr("contiguous")();
  py::array_t<float> arr = got.attr("numpy")().cast<py::array_t<float>>();
  std::memcpy(data, arr.data(), sizeof(float) * size);
}

void MV_AddMatrixTableAll(TableHandler handler, float* data, int size) {
  py::gil_scoped_acquire gil;
  auto& t = static_cast<Handle*>(handler)->table;
  t.attr("add")(to_tensor(data, size)).attr("wait")();
}

void MV_AddAsyncMatrixTableAll(TableHandler handler, float* data, int size) {
  py::gil_scoped_acquire gil;
  auto& t = static_cast<Handle*>(handler)->table;
  t.attr("add")(to_tensor(data, size).attr("clone")(),
                py::arg("async_op") = true);
}

static py::list row_list(int row_ids[], int n) {
  py::list rows;
  for (int i = 0; i < n; ++i) rows.append(row_ids[i]);
  return rows;
}

void MV_GetMatrixTableByRows(TableHandler handler, float* data, int size,
                             int row_ids[], int row_ids_n) {
  py::gil_scoped_acquire gil;
  auto& t = static_cast<Handle*>(handler)->table;
  py::object got = t.attr("get_rows")(row_list(row_ids, row_ids_n));
  got = got.attr("reshape")(-1).attr("cpu")().attr("contiguous")();
  py::array_t<float> arr = got.attr("numpy")().cast<py::array_t<float>>();
  std::memcpy(data, arr.data(), sizeof(float) * size);
}

void MV_AddMatrixTableByRows(TableHandler handler, float* data, int size,
                             int row_ids[], int row_ids_n) {
  py::gil_scoped_acquire gil;
  auto& t = static_cast<Handle*>(handler)->table;
  int num_col = t.attr("num_col").cast<int>();
  auto vals = to_tensor(data, size).attr("reshape")(
      py::make_tuple(row_ids_n, num_col));
  t.attr("add_rows")(row_list(row_ids, row_ids_n), vals);
}

void MV_AddAsyncMatrixTableByRows(TableHandler handler, float* data, int size,
                                  int row_ids[], int row_ids_n) {
  // keyed adds complete within the call in the collective design; the
  // async variant is equivalent (reference fire-and-forget semantics).
  MV_AddMatrixTableByRows(handler, data, size, row_ids, row_ids_n);
}

// ---- extensions beyond the reference C API (see c_api.h) ----

int MV_Rank() {
  py::gil_scoped_acquire gil;
  return mv().attr("rank")().cast<int>();
}

int MV_Size() {
  py::gil_scoped_acquire gil;
  return mv().attr("size")().cast<int>();
}

int MV_NumServers() {
  py::gil_scoped_acquire gil;
  return mv().attr("servers_num")().cast<int>();
}

void MV_Aggregate(float* data, int size) {
  py::gil_scoped_acquire gil;
  // in-place: aggregate() sums into the tensor view of caller memory
  mv().attr("aggregate")(to_tensor(data, size));
}

void MV_SetFlag(const char* key, const char* value) {
  ensure_interp();
  py::gil_scoped_acquire gil;
  mv().attr("set_flag")(std::string(key), std::string(value));
}

int MV_NetBind(int rank, const char* endpoint) {
  ensure_interp();
  py::gil_scoped_acquire gil;
  return mv().attr("net_bind")(rank, std::string(endpoint)).cast<bool>()
             ? 1 : 0;
}

int MV_NetConnect(int* ranks, char* endpoints[], int n) {
  ensure_interp();
  py::gil_scoped_acquire gil;
  py::list rs, eps;
  for (int i = 0; i < n; ++i) {
    rs.append(ranks[i]);
    eps.append(std::string(endpoints[i]));
  }
  return mv().attr("net_connect")(rs, eps).cast<bool>() ? 1 : 0;
}

}  // extern "C"
