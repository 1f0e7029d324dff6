"""Global auto-patch: PS-back every `torch.nn.Embedding` without touching
model code.

Parity with the reference's laboratory/inject demo
(laboratory/inject/openembedding_inject_tensorflow.py:1-38), which replaced
`tf.keras.layers.Embedding` process-wide via sitecustomize so unmodified
third-party models trained on the parameter server. Here `install()` swaps
`torch.nn.Embedding` for a factory returning the PS-backed layer when the
table is big enough (the same `sparse_as_dense_size` threshold policy as
`distributed_model`), and `uninstall()` restores torch.

    import openembedding_amd.inject as inject
    inject.install()                      # before model construction
    model = ThirdPartyCTRModel()          # its nn.Embedding(...) calls now
                                          # build openembedding_amd layers
"""

from __future__ import annotations

import torch
import torch.nn as nn

_original = nn.Embedding
_installed = False


def install(sparse_as_dense_size: int = 64) -> None:
    """Replace ``torch.nn.Embedding`` process-wide. Tables with fewer than
    ``sparse_as_dense_size`` rows stay ordinary dense modules (replicated,
    allreduce-trained — the reference's "cache" policy)."""
    global _installed
    if _installed:
        return

    def _factory(num_embeddings, embedding_dim, *args, **kwargs):
        if (not args and not kwargs
                and (num_embeddings < 0
                     or num_embeddings >= sparse_as_dense_size)):
            from .torch import Embedding
            return Embedding(num_embeddings, embedding_dim)
        # extra nn.Embedding options (padding_idx, sparse, ...) have no PS
        # equivalent — those tables stay plain torch
        return _original(num_embeddings, embedding_dim, *args, **kwargs)

    for mod in (nn, torch.nn.modules, torch.nn.modules.sparse):
        if getattr(mod, "Embedding", None) is _original:
            mod.Embedding = _factory
    _installed = True


def uninstall() -> None:
    global _installed
    for mod in (nn, torch.nn.modules, torch.nn.modules.sparse):
        if getattr(mod, "Embedding", None) is not _original:
            mod.Embedding = _original
    _installed = False
