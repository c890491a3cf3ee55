#!/usr/bin/env bash
# WordEmbedding launcher — the rebuild of the reference's example/run.bat
# (same knobs; enwiki-style corpus + optional precomputed vocab).
# Usage: ./examples/run_wordembedding.sh corpus.txt [vocab.txt] [ngpus]
set -e
cd "$(dirname "$0")/.."
TRAIN=${1:?usage: run_wordembedding.sh corpus.txt [vocab.txt] [ngpus]}
VOCAB=${2:-}
NGPUS=${3:-1}
SIZE=300; WINDOW=5; NEGATIVE=5; ALPHA=0.025; EPOCH=20; SAMPLE=0
MIN_COUNT=5; BLOCK=1000000; BINARY=1
ARGS=(-train_file "$TRAIN" -output embedding.bin -size $SIZE
      -window $WINDOW -negative $NEGATIVE -alpha $ALPHA -epoch $EPOCH
      -sample $SAMPLE -min_count $MIN_COUNT -data_block_size $BLOCK
      -binary $BINARY)
[ -n "$VOCAB" ] && ARGS+=(-read_vocab "$VOCAB")
if [ "$NGPUS" -gt 1 ]; then
  exec python -m torch.distributed.run --nnodes=1 --nproc-per-node "$NGPUS" \
      --master-addr 127.0.0.1 -m multiverso_amd.apps.wordembedding.main \
      "${ARGS[@]}"
fi
exec python -m multiverso_amd.apps.wordembedding.main "${ARGS[@]}"
