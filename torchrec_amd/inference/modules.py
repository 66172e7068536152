"""Inference build path: quantize + shard a trained model for serving.

Reference parity: torchrec/inference/modules.py (quantize_inference_model
:372, shard_quant_model :490, PredictFactory :189 / PredictModule :266).
"""

from __future__ import annotations

import abc
from typing import Any, Dict, List, Optional, Tuple, Type

import torch
import torch.nn as nn

from torchrec_amd.modules.embedding_modules import (
    EmbeddingBagCollection as FloatEBC,
    EmbeddingCollection as FloatEC,
)
from torchrec_amd.quant.embedding_modules import (
    EmbeddingBagCollection as QuantEBC,
    EmbeddingCollection as QuantEC,
)


def quantize_inference_model(
    model: nn.Module,
    quantization_mapping: Optional[Dict[Type[nn.Module], Any]] = None,
    output_dtype: torch.dtype = torch.float32,
) -> nn.Module:
    """Swap float embedding modules for int8 quantized ones in-place
    (reference inference/modules.py:372)."""
    mapping = quantization_mapping or {
        FloatEBC: QuantEBC,
        FloatEC: QuantEC,
    }

    def _swap(parent: nn.Module) -> None:
        for name, child in list(parent.named_children()):
            qcls = mapping.get(type(child))
            if qcls is not None:
                setattr(parent, name, qcls.from_float(child, output_dtype=output_dtype))
            else:
                _swap(child)

    _swap(model)
    return model


def shard_quant_model(
    model: nn.Module,
    world_size: int = 1,
    compute_device: str = "cuda",
    sharding_device: str = "cuda",
    constraints: Optional[Dict[str, Any]] = None,
) -> Tuple[nn.Module, Any]:
    """Place the quantized model for serving (reference inference/modules.py:490).

    Single-host inference: world_size==1 moves the quantized modules onto the
    target device. Multi-GPU single-host sharding (TW + all_to_one over xGMI)
    is the next milestone; the plan object records the placement decisions.
    """
    device = torch.device(compute_device if compute_device != "cuda" else "cuda:0")
    if world_size == 1:
        model = model.to(device)
        plan = {"world_size": 1, "placement": str(device)}
        return model, plan
    # multi-rank serving: shard quant EBCs with the training dists (TW/RW)
    import torch.distributed as dist

    from torchrec_amd.distributed.model_parallel import DistributedModelParallel
    from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner
    from torchrec_amd.distributed.planner.types import Topology
    from torchrec_amd.distributed.quant_embedding import (
        QuantEmbeddingCollectionSharder,
    )
    from torchrec_amd.distributed.quant_embeddingbag import (
        QuantEmbeddingBagCollectionSharder,
    )
    from torchrec_amd.distributed.types import ShardingEnv

    assert dist.is_initialized(), "multi-rank quant sharding needs torch.distributed"
    env = ShardingEnv.from_process_group(dist.group.WORLD)
    sharders = [QuantEmbeddingBagCollectionSharder(), QuantEmbeddingCollectionSharder()]
    planner = EmbeddingShardingPlanner(
        topology=Topology(
            world_size=world_size,
            compute_device=compute_device,
            hbm_cap=None if compute_device == "cuda" else 1 << 40,
        ),
        constraints=constraints,
    )
    plan = planner.collective_plan(model, sharders, dist.group.WORLD)
    dmp = DistributedModelParallel(
        model, env=env, plan=plan, sharders=sharders, device=device,
        init_data_parallel=False,
    )
    return dmp, plan


class PredictModule(nn.Module):
    """Serving wrapper contract (reference inference/modules.py:266):
    forward(batch dict) -> predictions dict."""

    def __init__(self, module: nn.Module) -> None:
        super().__init__()
        self._module = module
        self._module.eval()

    @property
    def predict_module(self) -> nn.Module:
        return self._module

    @abc.abstractmethod
    def predict_forward(self, batch: Dict[str, Any]) -> Any:
        ...

    def forward(self, batch: Dict[str, Any]) -> Any:
        with torch.inference_mode():
            return self.predict_forward(batch)


class PredictFactory(abc.ABC):
    """Packaging contract for serving artifacts (reference :189)."""

    @abc.abstractmethod
    def create_predict_module(self) -> nn.Module:
        ...

    def batching_metadata(self) -> Dict[str, str]:
        return {"float_features": "dense", "id_list_features": "sparse"}

    def result_metadata(self) -> str:
        return "dict_of_tensor"
