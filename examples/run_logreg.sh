#!/usr/bin/env bash
# LogisticRegression launcher — the rebuild of the reference's
# example/run.sh. Usage: ./examples/run_logreg.sh config_file [ngpus]
set -e
cd "$(dirname "$0")/.."
CFG=${1:?usage: run_logreg.sh config_file [ngpus]}
NGPUS=${2:-1}
if [ "$NGPUS" -gt 1 ]; then
  exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NGPUS" \
      --master-addr 127.0.0.1 -m multiverso_amd.apps.logreg.main "$CFG"
fi
exec python -m multiverso_amd.apps.logreg.main "$CFG"
