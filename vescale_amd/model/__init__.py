from .patch import VocabParallelCrossEntropy, VocabParallelEmbedding

__all__ = ["VocabParallelEmbedding", "VocabParallelCrossEntropy"]
