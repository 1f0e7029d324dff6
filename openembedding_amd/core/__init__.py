from .initializers import INITIALIZERS, Initializer, make_initializer
from .optimizers import OPTIMIZERS, SparseOptimizer, make_optimizer
from .variable import HASH_VOCAB_THRESHOLD, VariableMeta, VariableShard

__all__ = [
    "INITIALIZERS", "Initializer", "make_initializer",
    "OPTIMIZERS", "SparseOptimizer", "make_optimizer",
    "HASH_VOCAB_THRESHOLD", "VariableMeta", "VariableShard",
]
