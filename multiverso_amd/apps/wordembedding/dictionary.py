"""Word dictionary (word↔id↔frequency).

Capability parity with the reference Dictionary
(Applications/WordEmbedding/src/dictionary.h:34-68, 190 LoC): insert,
min_count filtering, id/word lookup, frequency access, and the vocab file
format used by -read_vocab / preprocess/word_count.cpp (word<space>count
per line)."""

from __future__ import annotations

from collections import Counter
from typing import Dict, Iterable, List, Optional


class Dictionary:
    def __init__(self) -> None:
        self.words: List[str] = []
        self.counts: List[int] = []
        self._index: Dict[str, int] = {}

    def __len__(self) -> int:
        return len(self.words)

    def insert(self, word: str, count: int = 1) -> int:
        idx = self._index.get(word)
        if idx is None:
            idx = len(self.words)
            self._index[word] = idx
            self.words.append(word)
            self.counts.append(count)
        else:
            self.counts[idx] += count
        return idx

    def get_id(self, word: str) -> int:
        return self._index.get(word, -1)

    def get_word(self, idx: int) -> str:
        return self.words[idx]

    def get_count(self, idx: int) -> int:
        return self.counts[idx]

    def remove_below(self, min_count: int) -> None:
        """min_count filtering (reference -min_count arg)."""
        kept = [(w, c) for w, c in zip(self.words, self.counts)
                if c >= min_count]
        self.words, self.counts, self._index = [], [], {}
        for w, c in kept:
            self.insert(w, c)

    @classmethod
    def build(cls, token_stream: Iterable[str],
              min_count: int = 5,
              stopwords: Optional[set] = None) -> "Dictionary":
        counter = Counter(token_stream)
        d = cls()
        # sort by descending count (word2vec convention; makes Huffman and
        # the unigram table cache-friendlier: hot rows cluster)
        for w, c in counter.most_common():
            if c >= min_count and (not stopwords or w not in stopwords):
                d.insert(w, c)
        return d

    # ---- vocab file io (word_count.cpp format) ----
    def save(self, path: str) -> None:
        with open(path, "w") as f:
            for w, c in zip(self.words, self.counts):
                f.write(f"{w} {c}\n")

    @classmethod
    def load(cls, path: str, min_count: int = 0) -> "Dictionary":
        d = cls()
        with open(path) as f:
            for line in f:
                parts = line.split()
                if len(parts) == 2 and int(parts[1]) >= min_count:
                    d.insert(parts[0], int(parts[1]))
        return d
