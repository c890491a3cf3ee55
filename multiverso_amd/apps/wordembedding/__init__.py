from .dictionary import Dictionary
from .huffman import HuffmanEncoder
from .model import WordEmbedding, WordEmbeddingOption
from .sampler import Sampler

__all__ = ["Dictionary", "HuffmanEncoder", "Sampler", "WordEmbedding",
           "WordEmbeddingOption"]
