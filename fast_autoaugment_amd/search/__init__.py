from .tpe import TPESampler, SearchSpace, ChoiceDim, UniformDim

__all__ = ["TPESampler", "SearchSpace", "ChoiceDim", "UniformDim"]
