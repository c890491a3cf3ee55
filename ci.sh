#!/usr/bin/env bash
# CI harness — the rebuild of the reference's Docker integration run
# (deploy/docker/Dockerfile:92-113: build, unit tests, binding tests,
# mpirun CLI tests). GPU tier runs only where an MI355X is visible.
set -e
cd "$(dirname "$0")"

echo "== build (gfx950 HIP extension + C API) =="
python __graft_entry__.py build
python -c 'from multiverso_amd import capi; capi.build(verbose=True)'

echo "== CPU suite (logic + single-process runtime + world_size 2/4 gloo \
+ ws3/ws8 + async-PS + binding + CLI end-to-end; GPU masked so device selection and \
multi-process semantics match the CPU tier) =="
HIP_VISIBLE_DEVICES="" CUDA_VISIBLE_DEVICES="" \
    python -m pytest tests -q -m "not gpu"

if python -c 'import torch,sys; sys.exit(0 if torch.cuda.is_available() else 1)'; then
  echo "== GPU suite (kernel numerics vs fp32 torch references, table ops, \
apps) =="
  python -m pytest tests -q -m gpu
  echo "== smoke =="
  python __graft_entry__.py smoke
fi
echo "CI OK"
