--- multiverso_amd Lua/Torch binding.
-- Capability parity with the reference binding/lua (init.lua:31-46,
-- ArrayTableHandler.lua, MatrixTableHandler.lua): LuaJIT FFI over the
-- C API (capi/c_api.h — the exact reference c_api.h symbol set), float
-- tensors only. Build libmultiverso_amd.so first
-- (python -c "from multiverso_amd import capi; capi.build()").
--
-- Usage:
--   local mv = require 'multiverso'
--   mv.init()
--   local tbl = mv.ArrayTableHandler:new(size)
--   tbl:add(delta); local t = tbl:get()
--   mv.barrier(); mv.shutdown()

local ffi = require 'ffi'

ffi.cdef[[
typedef void* TableHandler;
void MV_Init(int* argc, char* argv[]);
void MV_ShutDown();
void MV_Barrier();
int MV_NumWorkers();
int MV_WorkerId();
int MV_ServerId();
void MV_NewArrayTable(int size, TableHandler* out);
void MV_GetArrayTable(TableHandler handler, float* data, int size);
void MV_AddArrayTable(TableHandler handler, float* data, int size);
void MV_AddAsyncArrayTable(TableHandler handler, float* data, int size);
void MV_NewMatrixTable(int num_row, int num_col, TableHandler* out);
void MV_GetMatrixTableAll(TableHandler handler, float* data, int size);
void MV_AddMatrixTableAll(TableHandler handler, float* data, int size);
void MV_AddAsyncMatrixTableAll(TableHandler handler, float* data, int size);
void MV_GetMatrixTableByRows(TableHandler handler, float* data, int size,
                             int row_ids[], int row_ids_n);
void MV_AddMatrixTableByRows(TableHandler handler, float* data, int size,
                             int row_ids[], int row_ids_n);
int MV_Rank();
int MV_Size();
int MV_NumServers();
void MV_Aggregate(float* data, int size);
void MV_SetFlag(const char* key, const char* value);
void MV_AddAsyncMatrixTableByRows(TableHandler handler, float* data, int size,
                                  int row_ids[], int row_ids_n);
]]

local lib = ffi.load(os.getenv('MULTIVERSO_AMD_LIB') or 'multiverso_amd')

local mv = {}

function mv.init()
  local argc = ffi.new('int[1]', 0)
  lib.MV_Init(argc, nil)
end

function mv.shutdown() lib.MV_ShutDown() end
function mv.barrier() lib.MV_Barrier() end
function mv.rank() return lib.MV_Rank() end
function mv.size() return lib.MV_Size() end
function mv.servers_num() return lib.MV_NumServers() end
function mv.set_flag(k, v) lib.MV_SetFlag(k, tostring(v)) end
function mv.num_workers() return lib.MV_NumWorkers() end
function mv.worker_id() return lib.MV_WorkerId() end
function mv.server_id() return lib.MV_ServerId() end

-- torch.FloatTensor <-> C float* helpers (reference util.lua:17-31)
local function tensor_ptr(t)
  return t:data()
end

mv.ArrayTableHandler = {}
mv.ArrayTableHandler.__index = mv.ArrayTableHandler

function mv.ArrayTableHandler:new(size, init_value)
  local o = setmetatable({}, self)
  o.size = size
  local h = ffi.new('TableHandler[1]')
  lib.MV_NewArrayTable(size, h)
  o.handler = h[0]
  if init_value ~= nil then
    -- master adds the value, others add zeros (reference tables.py:50-57)
    local t = init_value:clone():float()
    if mv.worker_id() ~= 0 then t:zero() end
    o:add(t)
  end
  return o
end

function mv.ArrayTableHandler:get()
  local t = torch.FloatTensor(self.size)
  lib.MV_GetArrayTable(self.handler, tensor_ptr(t), self.size)
  return t
end

function mv.ArrayTableHandler:add(data, sync)
  local t = data:contiguous():float()
  if sync then
    lib.MV_AddArrayTable(self.handler, tensor_ptr(t), self.size)
  else
    lib.MV_AddAsyncArrayTable(self.handler, tensor_ptr(t), self.size)
  end
end

mv.MatrixTableHandler = {}
mv.MatrixTableHandler.__index = mv.MatrixTableHandler

function mv.MatrixTableHandler:new(num_row, num_col, init_value)
  local o = setmetatable({}, self)
  o.num_row, o.num_col = num_row, num_col
  local h = ffi.new('TableHandler[1]')
  lib.MV_NewMatrixTable(num_row, num_col, h)
  o.handler = h[0]
  if init_value ~= nil then
    local t = init_value:clone():float()
    if mv.worker_id() ~= 0 then t:zero() end
    o:add(t)
  end
  return o
end

function mv.MatrixTableHandler:get(row_ids)
  if row_ids == nil then
    local t = torch.FloatTensor(self.num_row, self.num_col)
    lib.MV_GetMatrixTableAll(self.handler, tensor_ptr(t),
                             self.num_row * self.num_col)
    return t
  end
  local n = #row_ids
  local ids = ffi.new('int[?]', n)
  for i = 1, n do ids[i - 1] = row_ids[i] end
  local t = torch.FloatTensor(n, self.num_col)
  lib.MV_GetMatrixTableByRows(self.handler, tensor_ptr(t),
                              n * self.num_col, ids, n)
  return t
end

function mv.MatrixTableHandler:add(data, row_ids, sync)
  local t = data:contiguous():float()
  if row_ids == nil then
    if sync then
      lib.MV_AddMatrixTableAll(self.handler, tensor_ptr(t),
                               self.num_row * self.num_col)
    else
      lib.MV_AddAsyncMatrixTableAll(self.handler, tensor_ptr(t),
                                    self.num_row * self.num_col)
    end
    return
  end
  local n = #row_ids
  local ids = ffi.new('int[?]', n)
  for i = 1, n do ids[i - 1] = row_ids[i] end
  if sync then
    lib.MV_AddMatrixTableByRows(self.handler, tensor_ptr(t),
                                n * self.num_col, ids, n)
  else
    lib.MV_AddAsyncMatrixTableByRows(self.handler, tensor_ptr(t),
                                     n * self.num_col, ids, n)
  end
end

return mv
