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

typedef void* TableHandler;

DllExport void MV_Init(int* argc, char* argv[]);
DllExport void MV_ShutDown();
DllExport void MV_Barrier();
DllExport int MV_NumWorkers();
DllExport int MV_WorkerId();
DllExport int MV_ServerId();

/* Array Table */
DllExport void MV_NewArrayTable(int size, TableHandler* out);
DllExport void MV_GetArrayTable(TableHandler handler, float* data, int size);
DllExport void MV_AddArrayTable(TableHandler handler, float* data, int size);
DllExport void MV_AddAsyncArrayTable(TableHandler handler, float* data,
                                     int size);

/* Matrix Table */
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

#ifdef __cplusplus
}
#endif

#endif /* MULTIVERSO_AMD_C_API_H_ */
