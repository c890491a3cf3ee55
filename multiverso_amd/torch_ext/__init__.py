from .param_manager import MVTorchParamManager
from .sharedvar import MVSharedTensor, mv_shared, sync_all_mv_shared

__all__ = ["MVTorchParamManager", "MVSharedTensor", "mv_shared",
           "sync_all_mv_shared"]
