/* multiverso_amd C API — symbol-for-symbol parity with the reference
 * include/multiverso/c_api.h:16-54 (float-only Array/Matrix table access,
 * runtime init/teardown/barrier, rank queries).
 *
 * The implementation (c_api.cpp) embeds the MI355X-native Python runtime:
 * a host process (C, Lua FFI, C#) linking this library drives the same
 * RCCL-over-xGMI collective data plane as the Python API.
 */
#ifndef MULTIVERSO_AMD_C_API_H_
#define MULTIVERSO_AMD_C_API_H_

#define DllExport

#ifdef __cplusplus
extern "C" {
#endif

/* Opaque table reference returned by the MV_New* constructors. The
 * handle owns a Python-side table object inside the embedded runtime;
 * it stays valid until MV_ShutDown. */
typedef void* TableHandler;

/* Runtime lifecycle. MV_Init boots the embedded MI355X runtime in this
 * process (rendezvous from the launcher environment, or from an argv
 * of "-key=value" flags such as -sync=true); MV_ShutDown drains pending
 * table work and tears the process group down; MV_Barrier blocks until
 * every rank arrives. */
DllExport void MV_Init(int* argc, char* argv[]);
DllExport void MV_ShutDown();
DllExport void MV_Barrier();

/* Topology queries: every rank is worker AND server in this design, so
 * NumWorkers == world size and WorkerId == ServerId == rank. */
DllExport int MV_NumWorkers();
DllExport int MV_WorkerId();
DllExport int MV_ServerId();

/* 1-D float table, contiguous-sharded across ranks. Get fills `data`
 * (length `size` floats) with the whole table via an all-gather of the
 * HBM-resident shards; Add applies `data` as a whole-table delta via a
 * reduce-scatter plus the server-side updater kernel. The Async add
 * returns once the payload is snapshotted; ordering against later Gets
 * is preserved. */
DllExport void MV_NewArrayTable(int size, TableHandler* out);
DllExport void MV_GetArrayTable(TableHandler handler, float* data, int size);
DllExport void MV_AddArrayTable(TableHandler handler, float* data, int size);
DllExport void MV_AddAsyncArrayTable(TableHandler handler, float* data,
                                     int size);

/* 2-D float table, row-sharded. The *All calls move the whole matrix
 * (size == num_row * num_col floats, row-major); the *ByRows calls move
 * the `row_ids_n` rows listed in `row_ids` (size == row_ids_n * num_col
 * floats), exchanged with their owning ranks as a batched all-to-all. */
DllExport void MV_NewMatrixTable(int num_row, int num_col, TableHandler* out);
DllExport void MV_GetMatrixTableAll(TableHandler handler, float* data,
                                    int size);
DllExport void MV_AddMatrixTableAll(TableHandler handler, float* data,
                                    int size);
DllExport void MV_AddAsyncMatrixTableAll(TableHandler handler, float* data,
                                         int size);
DllExport void MV_GetMatrixTableByRows(TableHandler handler, float* data,
                                       int size, int row_ids[], int row_ids_n);
DllExport void MV_AddMatrixTableByRows(TableHandler handler, float* data,
                                       int size, int row_ids[], int row_ids_n);
DllExport void MV_AddAsyncMatrixTableByRows(TableHandler handler, float* data,
                                            int size, int row_ids[],
                                            int row_ids_n);

/* ---- Extensions beyond the reference c_api.h (the reference exposed
 * these only through its C++ API, include/multiverso/multiverso.h:9-68;
 * they are provided here so the C++ header multiverso.hpp — and any
 * other FFI host — reaches the full MV_* surface). The exact reference
 * symbol set above is unchanged. ---- */

DllExport int MV_Rank();
DllExport int MV_Size();
DllExport int MV_NumServers();
/* In-place float sum-allreduce across all ranks (MV_Aggregate,
 * src/multiverso.cpp:53-56): rcclAllReduce over xGMI on GPU nodes. */
DllExport void MV_Aggregate(float* data, int size);
/* Programmatic flag set (MV_SetFlag, src/multiverso.cpp:48-51). */
DllExport void MV_SetFlag(const char* key, const char* value);
/* Explicit launcher-free rendezvous (MV_NetBind/MV_NetConnect,
 * src/multiverso.cpp:58-68): declare this process's rank, then provide
 * the full rank->endpoint map; call both BEFORE MV_Init. */
DllExport int MV_NetBind(int rank, const char* endpoint);
DllExport int MV_NetConnect(int* ranks, char* endpoints[], int n);

#ifdef __cplusplus
}
#endif

#endif /* MULTIVERSO_AMD_C_API_H_ */
