"""Distributed WordEmbedding CLI.

Capability parity with the reference binary
(Applications/WordEmbedding/src/main.cpp + argument parsing
util.cpp:31-57): same argument names (-size -train_file -read_vocab
-binary -cbow -alpha -output -window -sample -hs -negative -threads
-min_count -epoch -stopwords -sw_file -use_adagrad -data_block_size
-is_pipeline), word2vec-format output, per-epoch block pipeline with
word-count lr decay.

Launch one rank per GPU:
  python -m torch.distributed.run --nproc-per-node N \
      --master-addr 127.0.0.1 -m multiverso_amd.apps.wordembedding.main \
      -train_file corpus.txt -output emb.txt -size 200 -negative 5
"""

from __future__ import annotations

import argparse
import sys
import time

import torch

import multiverso_amd as mv

from .data import TextBlockReader, tokenize_file
from .dictionary import Dictionary
from .model import WordEmbedding, WordEmbeddingOption


def parse_args(argv):
    p = argparse.ArgumentParser(prefix_chars="-")
    p.add_argument("-size", type=int, default=100)
    p.add_argument("-train_file", type=str, required=True)
    p.add_argument("-read_vocab", type=str, default="")
    p.add_argument("-binary", type=int, default=0)
    p.add_argument("-cbow", type=int, default=0)
    p.add_argument("-alpha", type=float, default=0.025)
    p.add_argument("-output", type=str, default="embedding.txt")
    p.add_argument("-window", type=int, default=5)
    p.add_argument("-sample", type=float, default=0.0)
    p.add_argument("-hs", type=int, default=0)
    p.add_argument("-negative", type=int, default=5)
    p.add_argument("-threads", type=int, default=1,
                   help="kept for CLI parity; GPU waves replace threads")
    p.add_argument("-min_count", type=int, default=5)
    p.add_argument("-epoch", type=int, default=1)
    p.add_argument("-stopwords", type=int, default=0)
    p.add_argument("-sw_file", type=str, default="")
    p.add_argument("-use_adagrad", type=int, default=0)
    p.add_argument("-data_block_size", type=int, default=1_000_000)
    p.add_argument("-is_pipeline", type=int, default=1)
    return p.parse_args(argv)


def main(argv=None) -> None:
    args = parse_args(argv if argv is not None else sys.argv[1:])
    mv.init()

    stop = None
    if args.stopwords and args.sw_file:
        stop = set(open(args.sw_file).read().split())
    if args.read_vocab:
        # LoadVocab applies min_count after loading
        # (distributed_wordembedding.cpp:442)
        dictionary = Dictionary.load(args.read_vocab,
                                     min_count=args.min_count)
    else:
        dictionary = Dictionary.build(tokenize_file(args.train_file),
                                      min_count=args.min_count,
                                      stopwords=stop)
    total_words = sum(dictionary.counts)
    opt = WordEmbeddingOption(
        embedding_size=args.size, window=args.window,
        negative_num=args.negative, hs=bool(args.hs), cbow=bool(args.cbow),
        min_count=args.min_count, sample=args.sample,
        init_learning_rate=args.alpha, epoch=args.epoch,
        use_adagrad=bool(args.use_adagrad), total_words=total_words,
        data_block_size=args.data_block_size)
    model = WordEmbedding(opt, dictionary.counts)

    def block_stream(reader):
        """is_pipeline=1 (default, reference
        distributed_wordembedding.cpp:202-223): a loader thread tokenizes
        and stages block N+1 while block N trains on the GPU."""
        if not args.is_pipeline:
            yield from reader.blocks()
            return
        import queue
        import threading
        q: "queue.Queue" = queue.Queue(maxsize=2)

        def loader():
            for blk in reader.blocks():
                q.put(blk)
            q.put(None)

        t = threading.Thread(target=loader, daemon=True)
        t.start()
        while True:
            blk = q.get()
            if blk is None:
                break
            yield blk
        t.join()

    t0 = time.perf_counter()
    trained = 0
    empty = (torch.empty(0, dtype=torch.int64),
             torch.empty(0, dtype=torch.int64))
    for epoch in range(args.epoch):
        reader = TextBlockReader(args.train_file, dictionary,
                                 args.data_block_size, mv.rank(), mv.size())
        stream = block_stream(reader)
        while True:
            blk = next(stream, None)
            if mv.size() > 1:
                # ranks may hold different block counts (uneven corpus
                # stripes); train_block is collective, so every rank
                # keeps calling — with an empty block once exhausted —
                # until ALL ranks are done.
                have = torch.tensor([0.0 if blk is None else 1.0])
                mv.aggregate(have)
                if float(have[0]) == 0.0:
                    break
                if blk is None:
                    blk = empty
            elif blk is None:
                break
            trained += model.train_block(*blk)
            model.sync_word_count()
            if mv.rank() == 0:
                dt = time.perf_counter() - t0
                mv.log.info(
                    f"epoch {epoch} words {model.word_count_actual} "
                    f"lr {model.learning_rate:.5f} "
                    f"{model.word_count_actual / max(dt, 1e-9):.0f} words/s")
        mv.barrier()
    model.save_embedding(args.output, dictionary.words,
                         binary=bool(args.binary))
    mv.shutdown()


if __name__ == "__main__":
    main()
