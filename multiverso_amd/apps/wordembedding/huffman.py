"""Huffman encoder for hierarchical softmax.

Capability parity with the reference HuffmanEncoder
(Applications/WordEmbedding/src/huffman_encoder.h:15-58):
BuildFromTermFrequency produces, per word, the inner-node path (``point``)
and binary ``code`` arrays used by the HS output layer."""

from __future__ import annotations

from typing import List

import numpy as np


class HuffLabelInfo:
    __slots__ = ("point", "code")

    def __init__(self, point: List[int], code: List[int]) -> None:
        self.point = point  # inner-node ids along the root->word path
        self.code = code    # 0/1 branch labels


class HuffmanEncoder:
    def __init__(self) -> None:
        self.labels: List[HuffLabelInfo] = []

    def build_from_term_frequency(self, counts: List[int]) -> None:
        """Standard word2vec Huffman construction (two sorted queues)."""
        v = len(counts)
        if v == 0:
            self.labels = []
            return
        count = np.empty(2 * v - 1, dtype=np.int64)
        order = np.argsort(np.asarray(counts))[::-1]  # descending
        count[:v] = np.asarray(counts)[order]
        count[v:] = np.iinfo(np.int64).max
        parent = np.zeros(2 * v - 1, dtype=np.int64)
        binary = np.zeros(2 * v - 1, dtype=np.int8)
        pos1, pos2 = v - 1, v
        for a in range(v - 1):
            # pick two smallest
            picks = []
            for _ in range(2):
                if pos1 >= 0 and (pos2 >= 2 * v - 1 or
                                  count[pos1] < count[pos2]):
                    picks.append(pos1)
                    pos1 -= 1
                else:
                    picks.append(pos2)
                    pos2 += 1
            m1, m2 = picks
            count[v + a] = count[m1] + count[m2]
            parent[m1] = v + a
            parent[m2] = v + a
            binary[m2] = 1
        # trace paths
        labels_sorted: List[HuffLabelInfo] = []
        for a in range(v):
            code, point = [], []
            b = a
            while b != 2 * v - 2:
                code.append(int(binary[b]))
                point.append(int(parent[b] - v))
                b = int(parent[b])
            labels_sorted.append(HuffLabelInfo(point[::-1], code[::-1]))
        # undo the frequency sort: labels[original_word_id]
        self.labels = [None] * v  # type: ignore[list-item]
        for sorted_pos, orig in enumerate(order):
            self.labels[int(orig)] = labels_sorted[sorted_pos]

    def get_label_info(self, word_id: int) -> HuffLabelInfo:
        return self.labels[word_id]


    # ---- file io (huffman_encoder.cpp:9-85 format) ----
    def save_to_file(self, path: str, words: List[str]) -> None:
        """Save2File: header = vocab size; per line
        "word codelen code... point..."."""
        with open(path, "w") as f:
            f.write(f"{len(self.labels)}\n")
            for i, info in enumerate(self.labels):
                parts = [words[i], str(len(info.code))]
                parts += [str(c) for c in info.code]
                parts += [str(pt) for pt in info.point]
                f.write(" ".join(parts) + "\n")

    def load_from_file(self, path: str) -> List[str]:
        """RecoverFromFile: returns the word list; labels loaded in
        file order."""
        with open(path) as f:
            n = int(f.readline().split()[0])
            words: List[str] = []
            self.labels = []
            for _ in range(n):
                parts = f.readline().split()
                w, codelen = parts[0], int(parts[1])
                code = [int(x) for x in parts[2:2 + codelen]]
                point = [int(x) for x in parts[2 + codelen:2 + 2 * codelen]]
                words.append(w)
                self.labels.append(HuffLabelInfo(point, code))
        return words
