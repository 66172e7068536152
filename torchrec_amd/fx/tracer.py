"""torch.fx tracer treating sparse types and sharded modules as leaves.

Reference parity: torchrec/fx/tracer.py — lets models containing
KeyedJaggedTensor inputs / LazyAwaitable outputs be symbolically traced for
pipeline rewrites and export.
"""

from __future__ import annotations

from typing import Any, Dict, Optional

import torch
import torch.fx

from torchrec_amd.distributed.types import LazyAwaitable


class Tracer(torch.fx.Tracer):
    DEFAULT_LEAVES = [
        # branch on tensor properties (device/dtype) -> must stay opaque
        "InteractionArch",
        "InteractionDCNArch",
        "SparseArch",
    ]

    def __init__(self, leaf_modules: Optional[list] = None) -> None:
        super().__init__()
        self._leaf_modules = list(self.DEFAULT_LEAVES) + (leaf_modules or [])

    def is_leaf_module(self, m: torch.nn.Module, module_qualified_name: str) -> bool:
        if type(m).__name__ in self._leaf_modules:
            return True
        # sharded modules and TBEs are execution leaves
        if hasattr(m, "compute_and_output_dist") or hasattr(m, "split_embedding_weights"):
            return True
        from torchrec_amd.modules.embedding_modules import (
            EmbeddingBagCollection,
            EmbeddingCollection,
        )

        if isinstance(m, (EmbeddingBagCollection, EmbeddingCollection)):
            return True
        return super().is_leaf_module(m, module_qualified_name)

    def create_arg(self, a: Any):
        from torchrec_amd.sparse.jagged_tensor import (
            JaggedTensor,
            KeyedJaggedTensor,
            KeyedTensor,
        )

        if isinstance(a, (JaggedTensor, KeyedJaggedTensor, KeyedTensor, LazyAwaitable)):
            # opaque leaf values flow through the graph as constants
            return super().create_arg(a) if not isinstance(a, (JaggedTensor, KeyedJaggedTensor, KeyedTensor)) else a
        return super().create_arg(a)


def symbolic_trace(module: torch.nn.Module, leaf_modules: Optional[list] = None) -> torch.fx.GraphModule:
    tracer = Tracer(leaf_modules)
    graph = tracer.trace(module)
    return torch.fx.GraphModule(module, graph)
