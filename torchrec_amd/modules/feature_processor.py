"""Feature processors: per-position weighting before pooling.

Reference parity: torchrec/modules/feature_processor_.py
(FeatureProcessor :30, PositionWeightedModule :61,
PositionWeightedModuleCollection :198) and the FP-EBC wrapper
(torchrec/modules/fp_embedding_modules.py).
"""

from __future__ import annotations

import abc
from typing import Dict, List, Optional

import torch
import torch.nn as nn

from torchrec_amd import ops
from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection
from torchrec_amd.sparse.jagged_tensor import JaggedTensor, KeyedJaggedTensor, KeyedTensor


class FeatureProcessor(nn.Module, abc.ABC):
    """jt -> jt with weights (reference feature_processor_.py:30)."""

    @abc.abstractmethod
    def forward(self, features: JaggedTensor) -> JaggedTensor:
        ...


class PositionWeightedModule(FeatureProcessor):
    """Learnable weight per position within a bag (reference :61).

    Positions come from ops.lengths_range (the fbgemm offsets_range analogue).
    """

    def __init__(self, max_feature_length: int, device: Optional[torch.device] = None) -> None:
        super().__init__()
        self.position_weight = nn.Parameter(torch.ones(max_feature_length, device=device))

    def forward(self, features: JaggedTensor) -> JaggedTensor:
        positions = ops.lengths_range(features.offsets())
        weights = torch.gather(
            self.position_weight,
            0,
            positions.clamp(max=self.position_weight.numel() - 1),
        )
        return JaggedTensor(
            values=features.values(),
            lengths=features.lengths(),
            weights=weights,
        )


class PositionWeightedModuleCollection(nn.Module):
    """Per-feature position weights over a KJT (reference :198)."""

    def __init__(self, max_feature_lengths: Dict[str, int], device: Optional[torch.device] = None) -> None:
        super().__init__()
        self.max_feature_lengths = max_feature_lengths
        self.position_weights = nn.ParameterDict(
            {
                name: nn.Parameter(torch.ones(length, device=device))
                for name, length in max_feature_lengths.items()
            }
        )

    def forward(self, features: KeyedJaggedTensor) -> KeyedJaggedTensor:
        B = features.stride()
        positions = ops.lengths_range(features.offsets())
        opk = features.offset_per_key()
        weights = torch.ones_like(features.values(), dtype=torch.float32)
        for i, key in enumerate(features.keys()):
            if key not in self.position_weights:
                continue
            pw = self.position_weights[key]
            pos = positions[opk[i] : opk[i + 1]].clamp(max=pw.numel() - 1)
            weights[opk[i] : opk[i + 1]] = torch.gather(pw, 0, pos)
        return KeyedJaggedTensor(
            keys=features.keys(),
            values=features.values(),
            weights=weights,
            lengths=features.lengths(),
            stride=B,
        )


class FeatureProcessedEmbeddingBagCollection(nn.Module):
    """FP-EBC: apply processors, then a weighted EBC
    (reference modules/fp_embedding_modules.py)."""

    def __init__(
        self,
        embedding_bag_collection: EmbeddingBagCollection,
        feature_processors: nn.Module,  # PositionWeightedModuleCollection or dict
    ) -> None:
        super().__init__()
        assert embedding_bag_collection.is_weighted(), "FP-EBC needs a weighted EBC"
        self._embedding_bag_collection = embedding_bag_collection
        self._feature_processors = feature_processors

    def embedding_bag_configs(self):
        return self._embedding_bag_collection.embedding_bag_configs()

    def is_weighted(self) -> bool:
        return True

    def forward(self, features: KeyedJaggedTensor) -> KeyedTensor:
        if isinstance(self._feature_processors, PositionWeightedModuleCollection):
            processed = self._feature_processors(features)
        else:
            jts = features.to_dict()
            out: Dict[str, JaggedTensor] = {}
            for k, jt in jts.items():
                proc = (
                    self._feature_processors[k]
                    if k in self._feature_processors
                    else None
                )
                out[k] = proc(jt) if proc is not None else jt
            processed = KeyedJaggedTensor.from_jt_dict(out)
        return self._embedding_bag_collection(processed)
