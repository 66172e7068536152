"""torch.export-integrated IR: encapsulate embedding modules inside an
exported graph with meta-forward stubs, mark KJT inputs dynamic, and rebuild
the real modules from the serialized metadata after unflattening.

Reference parity: torchrec/ir/utils.py (ir_emb_lookup custom op :55,
encapsulate_ir_modules :135, decapsulate_ir_modules :166, mark_dynamic_kjt
:216) and torchrec/ir/serializer.py:94-161 (meta forwards).
"""

from __future__ import annotations

import types
from collections import defaultdict
from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn
from torch.export import Dim
from torch.export.dynamic_shapes import ShapesCollection

from torchrec_amd.ir.serializer import JsonSerializer
from torchrec_amd.modules.embedding_modules import (
    EmbeddingBagCollection,
    EmbeddingCollection,
)
from torchrec_amd.modules.feature_processor import FeatureProcessedEmbeddingBagCollection
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor, KeyedTensor


def _get_device(tensors: List[Optional[torch.Tensor]]) -> Optional[torch.device]:
    for t in tensors:
        if t is not None:
            return t.device
    return None


# meta-stub op: stands in for the embedding lookup inside the exported graph
# so torch.export never traces the real kernels (reference ir/utils.py:55)
@torch.library.custom_op("trec_amd_ir::ir_emb_lookup", mutates_args={})
def ir_emb_lookup_impl(
    tensors: List[Optional[torch.Tensor]], batch_size: int, dims: List[int]
) -> List[torch.Tensor]:
    device = _get_device(tensors)
    return [torch.empty(batch_size, dim, device=device) for dim in dims]


@torch.library.register_fake("trec_amd_ir::ir_emb_lookup")
def ir_emb_lookup_fake(
    tensors: List[Optional[torch.Tensor]], batch_size: int, dims: List[int]
) -> List[torch.Tensor]:
    device = _get_device(tensors)
    return [torch.empty(batch_size, dim, device=device) for dim in dims]


def _ebc_meta_forward(ebc, features: KeyedJaggedTensor) -> KeyedTensor:
    batch_size = features.stride()
    arg_list = [
        features.values(),
        features.weights_or_none(),
        features.lengths_or_none(),
        features.offsets_or_none(),
    ]
    dims = list(ebc._lengths_per_embedding)
    outputs = torch.ops.trec_amd_ir.ir_emb_lookup(arg_list, batch_size, dims)
    return KeyedTensor(
        keys=list(ebc._feature_names),
        values=torch.cat(outputs, dim=1),
        length_per_key=dims,
    )


def _fpebc_meta_forward(fpebc, features: KeyedJaggedTensor) -> KeyedTensor:
    ebc = fpebc._embedding_bag_collection
    return _ebc_meta_forward(ebc, features)


_KJT_PYTREE_REGISTERED = False


def register_kjt_pytree() -> None:
    """Register KeyedJaggedTensor as a pytree node so torch.export accepts it
    as a (flattened) graph input. Idempotent."""
    global _KJT_PYTREE_REGISTERED
    if _KJT_PYTREE_REGISTERED:
        return

    # context carries only the keys: stride derives from len(lengths)/len(keys)
    # so the batch dimension stays dynamic across unflattened calls
    def _flatten(kjt: KeyedJaggedTensor):
        return (
            [kjt.values(), kjt.lengths(), kjt.weights_or_none()],
            list(kjt.keys()),
        )

    def _unflatten(values, context):
        vals, lengths, weights = values
        return KeyedJaggedTensor(
            keys=context, values=vals, lengths=lengths, weights=weights
        )

    def _flatten_spec(kjt, spec):
        return [kjt.values(), kjt.lengths(), kjt.weights_or_none()]

    def _flatten_with_keys(kjt: KeyedJaggedTensor):
        values, context = _flatten(kjt)
        names = ["values", "lengths", "weights"]
        return [
            (torch.utils._pytree.GetAttrKey(n), v) for n, v in zip(names, values)
        ], context

    torch.utils._pytree.register_pytree_node(
        KeyedJaggedTensor, _flatten, _unflatten,
        serialized_type_name="torchrec_amd.sparse.jagged_tensor.KeyedJaggedTensor",
        flatten_with_keys_fn=_flatten_with_keys,
    )
    try:
        from torch.fx._pytree import register_pytree_flatten_spec

        register_pytree_flatten_spec(KeyedJaggedTensor, _flatten_spec)
    except Exception:
        pass

    # KeyedTensor flows OUT of preserved module-call signatures
    def _kt_flatten(kt: KeyedTensor):
        return [kt.values()], (list(kt.keys()), list(kt.length_per_key()))

    def _kt_unflatten(values, context):
        return KeyedTensor(
            keys=context[0], values=values[0], length_per_key=context[1]
        )

    def _kt_flatten_with_keys(kt: KeyedTensor):
        values, context = _kt_flatten(kt)
        return [(torch.utils._pytree.GetAttrKey("values"), values[0])], context

    torch.utils._pytree.register_pytree_node(
        KeyedTensor, _kt_flatten, _kt_unflatten,
        serialized_type_name="torchrec_amd.sparse.jagged_tensor.KeyedTensor",
        flatten_with_keys_fn=_kt_flatten_with_keys,
    )
    try:
        from torch.fx._pytree import register_pytree_flatten_spec

        register_pytree_flatten_spec(KeyedTensor, lambda kt, spec: [kt.values()])
    except Exception:
        pass
    _KJT_PYTREE_REGISTERED = True


_META_FORWARDS = {
    EmbeddingBagCollection: _ebc_meta_forward,
    FeatureProcessedEmbeddingBagCollection: _fpebc_meta_forward,
}


def encapsulate_ir_modules(
    module: nn.Module, fqn: str = ""
) -> Tuple[nn.Module, List[str]]:
    """Serialize each embedding module's config into an ``ir_metadata`` uint8
    buffer and swap its forward for the meta stub, so torch.export records a
    single opaque ir_emb_lookup in its place (reference ir/utils.py:135)."""
    register_kjt_pytree()
    preserved: List[str] = []
    for child_fqn, child in module.named_modules():
        meta_fwd = None
        for cls, fwd in _META_FORWARDS.items():
            if type(child) is cls:
                meta_fwd = fwd
                break
        if meta_fwd is None:
            continue
        blob, _ = JsonSerializer.serialize(child)
        child.register_buffer(
            "ir_metadata",
            torch.frombuffer(bytearray(blob), dtype=torch.uint8).clone(),
            persistent=True,
        )
        child.forward = types.MethodType(meta_fwd, child)
        preserved.append(child_fqn if not fqn else f"{fqn}.{child_fqn}")
    return module, preserved


def decapsulate_ir_modules(
    module: nn.Module, device: Optional[torch.device] = None
) -> nn.Module:
    """Rebuild the real embedding modules from ``ir_metadata`` buffers inside
    an unflattened exported module (reference ir/utils.py:166)."""
    for child_fqn, child in list(module.named_children()):
        child = decapsulate_ir_modules(child, device)
        setattr(module, child_fqn, child)
    buffers = dict(module.named_buffers(recurse=False))
    if "ir_metadata" in buffers:
        blob = bytes(buffers["ir_metadata"].cpu().numpy().tobytes())
        module = JsonSerializer.deserialize(blob, device)
    return module


_DYNAMIC_DIMS: Dict[str, int] = defaultdict(int)


def _get_dim(name: str, min: Optional[int] = None, max: Optional[int] = None):
    dim = f"{name}_{_DYNAMIC_DIMS[name]}"
    _DYNAMIC_DIMS[name] += 1
    return Dim(dim, min=min, max=max)


def mark_dynamic_kjt(
    kjt: KeyedJaggedTensor,
    shapes_collection: Optional[ShapesCollection] = None,
    variable_length: bool = False,
    variable_batch: bool = False,
    vlen=None,
    llen=None,
) -> ShapesCollection:
    """Mark the KJT's flattened tensors dynamic for torch.export (reference
    ir/utils.py:216): values (and weights) share one dynamic dim ``vlen``;
    with ``variable_length`` the lengths get their own dim ``llen``."""
    register_kjt_pytree()
    if shapes_collection is None:
        shapes_collection = ShapesCollection()
    # min=2: empty KJTs are padded, and observed-size guards need headroom
    vlen = _get_dim("vlen", min=2) if vlen is None else vlen
    if kjt._values is not None and kjt._values.dim() > 0:
        shapes_collection[kjt._values] = (vlen,)
    w = kjt.weights_or_none()
    if w is not None and w.dim() > 0:
        shapes_collection[w] = (vlen,)
    if variable_length or variable_batch:
        # variable_batch: len(lengths) == len(keys) * B with B dynamic;
        # variable_length: per-feature batch sizes differ too
        # min = 2 keys' worth: keeps batch >= 2 so B==1 broadcast guards
        # never specialize the graph
        llen = (
            _get_dim("llen", min=max(2, 2 * len(kjt.keys())))
            if llen is None
            else llen
        )
        lengths = kjt.lengths_or_none()
        if lengths is not None and lengths.dim() > 0:
            shapes_collection[lengths] = (llen,)
    return shapes_collection
