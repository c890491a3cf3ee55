// C# binding for multiverso_amd — the MI355X-native equivalent of the
// reference's C++/CLI MultiversoCLR wrapper (binding/C#/MultiversoCLR/
// MultiversoCLR.h:12-45). C++/CLI is Windows-only; on a ROCm/Linux node
// the idiomatic surface is pure C# P/Invoke over the same C API
// (libmultiverso_amd.so, capi/c_api.h) that the Lua binding uses, so the
// capability set matches MultiversoWrapper: Init/Shutdown/Barrier/rank
// queries, table creation, generic whole-table and by-row Get/Add.
//
// The C API is float-only (reference c_api.h:16-54); the generic Get/Add
// of the CLR wrapper likewise routed every element type through float
// tables (MatrixTable.h:10-80), so float[] is the payload type here.
//
// Build:  dotnet build  (or: csc -t:library Multiverso.cs)
// Run:    LD_LIBRARY_PATH must include the directory holding
//         libmultiverso_amd.so (built by multiverso_amd/capi/build.py).

using System;
using System.Runtime.InteropServices;

namespace Multiverso
{
    public static class MultiversoWrapper
    {
        private const string Lib = "multiverso_amd";

        // ---- raw C API ------------------------------------------------
        [DllImport(Lib, EntryPoint = "MV_Init")]
        private static extern void MV_Init(ref int argc, string[] argv);
        [DllImport(Lib, EntryPoint = "MV_ShutDown")]
        private static extern void MV_ShutDown();
        [DllImport(Lib, EntryPoint = "MV_Barrier")]
        private static extern void MV_Barrier();
        [DllImport(Lib, EntryPoint = "MV_NumWorkers")]
        private static extern int MV_NumWorkers();
        [DllImport(Lib, EntryPoint = "MV_WorkerId")]
        private static extern int MV_WorkerId();
        [DllImport(Lib, EntryPoint = "MV_ServerId")]
        private static extern int MV_ServerId();

        [DllImport(Lib, EntryPoint = "MV_Rank")]
        public static extern int Rank();

        [DllImport(Lib, EntryPoint = "MV_Size")]
        public static extern int Size();

        [DllImport(Lib, EntryPoint = "MV_NumServers")]
        public static extern int NumServers();

        [DllImport(Lib, EntryPoint = "MV_Aggregate")]
        public static extern void Aggregate(float[] data, int size);

        [DllImport(Lib, EntryPoint = "MV_SetFlag")]
        public static extern void SetFlag(string key, string value);

        [DllImport(Lib, EntryPoint = "MV_NewArrayTable")]
        private static extern void MV_NewArrayTable(int size, out IntPtr handler);
        [DllImport(Lib, EntryPoint = "MV_GetArrayTable")]
        private static extern void MV_GetArrayTable(IntPtr handler, float[] data, int size);
        [DllImport(Lib, EntryPoint = "MV_AddArrayTable")]
        private static extern void MV_AddArrayTable(IntPtr handler, float[] data, int size);
        [DllImport(Lib, EntryPoint = "MV_AddAsyncArrayTable")]
        private static extern void MV_AddAsyncArrayTable(IntPtr handler, float[] data, int size);

        [DllImport(Lib, EntryPoint = "MV_NewMatrixTable")]
        private static extern void MV_NewMatrixTable(int rows, int cols, out IntPtr handler);
        [DllImport(Lib, EntryPoint = "MV_GetMatrixTableAll")]
        private static extern void MV_GetMatrixTableAll(IntPtr handler, float[] data, int size);
        [DllImport(Lib, EntryPoint = "MV_AddMatrixTableAll")]
        private static extern void MV_AddMatrixTableAll(IntPtr handler, float[] data, int size);
        [DllImport(Lib, EntryPoint = "MV_AddAsyncMatrixTableAll")]
        private static extern void MV_AddAsyncMatrixTableAll(IntPtr handler, float[] data, int size);
        [DllImport(Lib, EntryPoint = "MV_GetMatrixTableByRows")]
        private static extern void MV_GetMatrixTableByRows(IntPtr handler, float[] data,
                                                           int size, int[] rowIds, int n);
        [DllImport(Lib, EntryPoint = "MV_AddMatrixTableByRows")]
        private static extern void MV_AddMatrixTableByRows(IntPtr handler, float[] data,
                                                           int size, int[] rowIds, int n);
        [DllImport(Lib, EntryPoint = "MV_AddAsyncMatrixTableByRows")]
        private static extern void MV_AddAsyncMatrixTableByRows(IntPtr handler, float[] data,
                                                                int size, int[] rowIds, int n);

        // ---- MultiversoWrapper surface (MultiversoCLR.h parity) -------
        private static IntPtr[] _tables = Array.Empty<IntPtr>();
        private static int[] _cols = Array.Empty<int>();

        /// Init(num_tables, sync): sync mode is selected with the same
        /// -sync=true argv convention the Python binding uses
        /// (reference api.py:30-34).
        public static void Init(int numTables, bool sync)
        {
            var argv = sync ? new[] { "csharp", "-sync=true" } : new[] { "csharp" };
            int argc = argv.Length;
            MV_Init(ref argc, argv);
            _tables = new IntPtr[numTables];
            _cols = new int[numTables];
        }

        public static void Shutdown() => MV_ShutDown();
        public static int Rank() => MV_WorkerId();
        public static int Size() => MV_NumWorkers();
        public static void Barrier() => MV_Barrier();

        /// CreateTable(table_id, rows, cols): cols == 1 creates an
        /// ArrayTable of `rows` elements, otherwise a MatrixTable.
        public static void CreateTable(int tableId, int rows, int cols)
        {
            if (cols <= 1) { MV_NewArrayTable(rows, out _tables[tableId]); _cols[tableId] = 1; }
            else { MV_NewMatrixTable(rows, cols, out _tables[tableId]); _cols[tableId] = cols; }
        }

        public static void CreateTables(int[] rows, int[] cols)
        {
            for (int i = 0; i < rows.Length; ++i) CreateTable(i, rows[i], cols[i]);
        }

        public static void Get(int tableId, float[] value)
        {
            if (_cols[tableId] == 1) MV_GetArrayTable(_tables[tableId], value, value.Length);
            else MV_GetMatrixTableAll(_tables[tableId], value, value.Length);
        }

        public static void Get(int tableId, int rowId, float[] value)
        {
            MV_GetMatrixTableByRows(_tables[tableId], value, value.Length,
                                    new[] { rowId }, 1);
        }

        public static void GetRows(int tableId, int[] rowIds, float[] value)
        {
            MV_GetMatrixTableByRows(_tables[tableId], value, value.Length,
                                    rowIds, rowIds.Length);
        }

        public static void Add(int tableId, float[] update, bool async = false)
        {
            var h = _tables[tableId];
            if (_cols[tableId] == 1)
            {
                if (async) MV_AddAsyncArrayTable(h, update, update.Length);
                else MV_AddArrayTable(h, update, update.Length);
            }
            else
            {
                if (async) MV_AddAsyncMatrixTableAll(h, update, update.Length);
                else MV_AddMatrixTableAll(h, update, update.Length);
            }
        }

        public static void Add(int tableId, int rowId, float[] value, bool async = false)
        {
            var ids = new[] { rowId };
            if (async) MV_AddAsyncMatrixTableByRows(_tables[tableId], value, value.Length, ids, 1);
            else MV_AddMatrixTableByRows(_tables[tableId], value, value.Length, ids, 1);
        }
    }
}
