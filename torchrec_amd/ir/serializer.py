"""IR export: serialize embedding-module configs for model export.

Reference parity: torchrec/ir/serializer.py:94-161 (JSON config
serialization with meta-forward stubs) and torchrec/ir/utils.py
(encapsulate_ir_modules :135 / decapsulate_ir_modules :166,
mark_dynamic_kjt :216).
"""

from __future__ import annotations

import json
from dataclasses import asdict
from typing import Dict, List, Optional, Tuple, Type

import torch
import torch.nn as nn

from torchrec_amd.modules.embedding_configs import (
    DataType,
    EmbeddingBagConfig,
    EmbeddingConfig,
    PoolingType,
)
from torchrec_amd.modules.embedding_modules import (
    EmbeddingBagCollection,
    EmbeddingCollection,
)


def _config_to_dict(cfg) -> Dict:
    d = asdict(cfg)
    d["data_type"] = cfg.data_type.value
    if hasattr(cfg, "pooling"):
        d["pooling"] = cfg.pooling.value
    return d


def _dict_to_bag_config(d: Dict) -> EmbeddingBagConfig:
    d = dict(d)
    d["data_type"] = DataType(d["data_type"])
    d["pooling"] = PoolingType(d.get("pooling", "SUM"))
    d.pop("weight_init_max", None) if d.get("weight_init_max") is None else None
    return EmbeddingBagConfig(**d)


def _dict_to_config(d: Dict) -> EmbeddingConfig:
    d = dict(d)
    d["data_type"] = DataType(d["data_type"])
    d.pop("pooling", None)
    return EmbeddingConfig(**d)


class JsonSerializer:
    """Module <-> JSON metadata (reference ir/serializer.py:161)."""

    @staticmethod
    def serialize(module: nn.Module) -> Tuple[bytes, str]:
        if isinstance(module, EmbeddingBagCollection):
            payload = {
                "type": "EmbeddingBagCollection",
                "tables": [_config_to_dict(c) for c in module.embedding_bag_configs()],
                "is_weighted": module.is_weighted(),
            }
        elif isinstance(module, EmbeddingCollection):
            payload = {
                "type": "EmbeddingCollection",
                "tables": [_config_to_dict(c) for c in module.embedding_configs()],
                "need_indices": module.need_indices(),
            }
        else:
            from torchrec_amd.modules.feature_processor import (
                FeatureProcessedEmbeddingBagCollection,
                PositionWeightedModuleCollection,
            )

            if isinstance(module, FeatureProcessedEmbeddingBagCollection):
                fp = module._feature_processors
                if not isinstance(fp, PositionWeightedModuleCollection):
                    raise NotImplementedError(
                        "only PositionWeightedModuleCollection FP-EBCs serialize"
                    )
                payload = {
                    "type": "FeatureProcessedEmbeddingBagCollection",
                    "tables": [
                        _config_to_dict(c) for c in module.embedding_bag_configs()
                    ],
                    "max_feature_lengths": dict(fp.max_feature_lengths),
                }
            else:
                raise NotImplementedError(f"cannot serialize {type(module)}")
        return json.dumps(payload).encode(), payload["type"]

    @staticmethod
    def deserialize(data: bytes, device: Optional[torch.device] = None) -> nn.Module:
        payload = json.loads(data.decode())
        if payload["type"] == "EmbeddingBagCollection":
            return EmbeddingBagCollection(
                tables=[_dict_to_bag_config(d) for d in payload["tables"]],
                is_weighted=payload["is_weighted"],
                device=device,
            )
        if payload["type"] == "EmbeddingCollection":
            return EmbeddingCollection(
                tables=[_dict_to_config(d) for d in payload["tables"]],
                need_indices=payload["need_indices"],
                device=device,
            )
        if payload["type"] == "FeatureProcessedEmbeddingBagCollection":
            from torchrec_amd.modules.feature_processor import (
                FeatureProcessedEmbeddingBagCollection,
                PositionWeightedModuleCollection,
            )

            ebc = EmbeddingBagCollection(
                tables=[_dict_to_bag_config(d) for d in payload["tables"]],
                is_weighted=True,
                device=device,
            )
            return FeatureProcessedEmbeddingBagCollection(
                ebc,
                PositionWeightedModuleCollection(
                    {k: int(v) for k, v in payload["max_feature_lengths"].items()},
                    device=device,
                ),
            )
        raise NotImplementedError(payload["type"])


def encapsulate_ir_modules(model: nn.Module) -> Dict[str, bytes]:
    """Record serialized metadata for every embedding module in the tree
    (reference ir/utils.py:135)."""
    out: Dict[str, bytes] = {}
    for fqn, child in model.named_modules():
        if isinstance(child, (EmbeddingBagCollection, EmbeddingCollection)):
            out[fqn], _ = JsonSerializer.serialize(child)
    return out


def decapsulate_ir_modules(
    model: nn.Module, metadata: Dict[str, bytes], device: Optional[torch.device] = None
) -> nn.Module:
    """Rebuild embedding modules from serialized metadata (reference :166)."""
    for fqn, blob in metadata.items():
        rebuilt = JsonSerializer.deserialize(blob, device)
        parent = model
        parts = fqn.split(".")
        for p in parts[:-1]:
            parent = getattr(parent, p)
        setattr(parent, parts[-1], rebuilt)
    return model
