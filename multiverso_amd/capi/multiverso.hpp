// multiverso_amd C++ API — the reference's public C++ surface
// (include/multiverso/multiverso.h:9-68 free functions + the worker
// table handler classes of binding use) for native C++ hosts.
//
// Header-only over the C API (c_api.h): a C++ application links
// libmultiverso_amd.so and drives the same MI355X-native runtime
// (RCCL-over-xGMI collective plane in sync mode; host-lane async PS in
// the default mode) as the Python API. Tables are float-only at this
// surface, matching the reference C API (c_api.h:16-54); the reference's
// templated MV_CreateTable<Option> reduces to the two typed handlers
// below (its int/double instantiations are reachable via the Python
// API's dtype argument).

#ifndef MULTIVERSO_AMD_MULTIVERSO_HPP_
#define MULTIVERSO_AMD_MULTIVERSO_HPP_

#include <cstddef>
#include <stdexcept>
#include <string>
#include <vector>

#include "c_api.h"

namespace multiverso {

// ---- runtime lifecycle (multiverso.h:9-33) ----
inline void Init(int* argc = nullptr, char** argv = nullptr) {
  MV_Init(argc, argv);
}
inline void ShutDown() { MV_ShutDown(); }
inline void Barrier() { MV_Barrier(); }

// ---- topology (multiverso.h:21-31) ----
inline int Rank() { return MV_Rank(); }
inline int Size() { return MV_Size(); }
inline int NumWorkers() { return MV_NumWorkers(); }
inline int NumServers() { return MV_NumServers(); }
inline int WorkerId() { return MV_WorkerId(); }
inline int ServerId() { return MV_ServerId(); }

// ---- model-average aggregate (multiverso.h:43-48) ----
inline void Aggregate(float* data, int size) { MV_Aggregate(data, size); }
inline void Aggregate(std::vector<float>& data) {
  MV_Aggregate(data.data(), static_cast<int>(data.size()));
}

// ---- flags + explicit rendezvous (multiverso.h:50-68) ----
inline void SetCMDFlag(const std::string& key, const std::string& value) {
  MV_SetFlag(key.c_str(), value.c_str());
}
inline bool NetBind(int rank, const std::string& endpoint) {
  return MV_NetBind(rank, endpoint.c_str()) != 0;
}
inline bool NetConnect(const std::vector<int>& ranks,
                       const std::vector<std::string>& endpoints) {
  std::vector<int> rs(ranks);
  std::vector<char*> eps;
  eps.reserve(endpoints.size());
  for (const auto& e : endpoints) eps.push_back(const_cast<char*>(e.c_str()));
  return MV_NetConnect(rs.data(), eps.data(),
                       static_cast<int>(rs.size())) != 0;
}

// ---- table handlers (the reference WorkerTable client surface,
// table_interface.h:24-56, as the binding-style handler pair) ----

class ArrayTableHandler {
 public:
  explicit ArrayTableHandler(int size) : size_(size) {
    MV_NewArrayTable(size, &handle_);
  }
  // handle lifetime: owned by the runtime until ShutDown (c_api.h)
  int size() const { return size_; }
  void Get(float* data) { MV_GetArrayTable(handle_, data, size_); }
  void Get(std::vector<float>& data) {
    data.resize(size_);
    Get(data.data());
  }
  void Add(const float* delta) {
    MV_AddArrayTable(handle_, const_cast<float*>(delta), size_);
  }
  void AddAsync(const float* delta) {
    MV_AddAsyncArrayTable(handle_, const_cast<float*>(delta), size_);
  }

 private:
  TableHandler handle_ = nullptr;
  int size_;
};

class MatrixTableHandler {
 public:
  MatrixTableHandler(int num_row, int num_col)
      : rows_(num_row), cols_(num_col) {
    MV_NewMatrixTable(num_row, num_col, &handle_);
  }
  int num_row() const { return rows_; }
  int num_col() const { return cols_; }

  void GetAll(float* data) {
    MV_GetMatrixTableAll(handle_, data, rows_ * cols_);
  }
  void AddAll(const float* delta) {
    MV_AddMatrixTableAll(handle_, const_cast<float*>(delta), rows_ * cols_);
  }
  void AddAsyncAll(const float* delta) {
    MV_AddAsyncMatrixTableAll(handle_, const_cast<float*>(delta),
                              rows_ * cols_);
  }
  void GetByRows(float* data, const std::vector<int>& row_ids) {
    MV_GetMatrixTableByRows(handle_, data,
                            static_cast<int>(row_ids.size()) * cols_,
                            const_cast<int*>(row_ids.data()),
                            static_cast<int>(row_ids.size()));
  }
  void AddByRows(const float* delta, const std::vector<int>& row_ids) {
    MV_AddMatrixTableByRows(handle_, const_cast<float*>(delta),
                            static_cast<int>(row_ids.size()) * cols_,
                            const_cast<int*>(row_ids.data()),
                            static_cast<int>(row_ids.size()));
  }
  void AddAsyncByRows(const float* delta, const std::vector<int>& row_ids) {
    MV_AddAsyncMatrixTableByRows(handle_, const_cast<float*>(delta),
                                 static_cast<int>(row_ids.size()) * cols_,
                                 const_cast<int*>(row_ids.data()),
                                 static_cast<int>(row_ids.size()));
  }

 private:
  TableHandler handle_ = nullptr;
  int rows_, cols_;
};

}  // namespace multiverso

#endif  // MULTIVERSO_AMD_MULTIVERSO_HPP_
