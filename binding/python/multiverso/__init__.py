"""Drop-in `multiverso` package — the reference binding's import name
(binding/python/multiverso: api.py + tables.py), re-exported from the
MI355X-native implementation so reference training scripts run
unchanged:

    import multiverso as mv
    mv.init(sync=True)
    tbl = mv.ArrayTableHandler(1000, init_value=w0)
    ...
    mv.barrier(); mv.shutdown()

Install by adding this directory's parent to PYTHONPATH (or pip-install
the repo root; setup.py maps both packages). The theano/lasagne/keras
extensions of the reference (theano_ext.sharedvar, *_ext.param_manager)
are superseded by `multiverso_amd.torch_ext` (`MVSharedTensor`,
`MVTorchParamManager`) — same delta-merge protocol
(sharedvar.py:37-44 / param_manager.py:67-81) on torch tensors.
"""

from multiverso_amd import (  # noqa: F401
    init, shutdown, barrier, net_bind, net_connect,
    workers_num, servers_num, worker_id, server_id, is_master_worker,
    rank, size, aggregate,
    ArrayTableHandler, MatrixTableHandler,
    ArrayTable, MatrixTable, SparseMatrixTable, KVTable,
    AddOption, set_flag, get_flag,
    checkpoint, restore,
)

from multiverso_amd import torch_ext  # noqa: F401
