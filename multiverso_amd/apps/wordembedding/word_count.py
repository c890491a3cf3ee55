"""Vocabulary builder CLI — parity with the reference preprocessing tool
(Applications/WordEmbedding/preprocess/word_count.cpp): counts words in a
corpus and writes the vocab file consumed by -read_vocab.

Usage: python -m multiverso_amd.apps.wordembedding.word_count \
           <corpus.txt> <vocab.txt> [min_count]
"""

import sys

from .data import tokenize_file
from .dictionary import Dictionary


def main() -> None:
    if len(sys.argv) < 3:
        print(__doc__)
        sys.exit(1)
    corpus, out = sys.argv[1], sys.argv[2]
    min_count = int(sys.argv[3]) if len(sys.argv) > 3 else 5
    d = Dictionary.build(tokenize_file(corpus), min_count=min_count)
    d.save(out)
    print(f"{len(d)} words (min_count={min_count}) -> {out}")


if __name__ == "__main__":
    main()
