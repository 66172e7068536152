"""Two-tower retrieval (BASELINE config #5): train + quantized inference."""

import torch

from torchrec_amd.inference.modules import quantize_inference_model
from torchrec_amd.models.two_tower import SequenceTwoTower, TwoTower, TwoTowerTrain
from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig, EmbeddingConfig
from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection, EmbeddingCollection
from torchrec_amd.quant.embedding_modules import EmbeddingBagCollection as QuantEBC
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


def _kjt(B=4):
    torch.manual_seed(0)
    lengths = torch.randint(1, 4, (2 * B,))
    values = torch.randint(0, 50, (int(lengths.sum()),))
    return KeyedJaggedTensor(
        keys=["user", "item"], values=values, lengths=lengths, stride=B
    )


def _tables():
    return [
        EmbeddingBagConfig(num_embeddings=50, embedding_dim=16, name="t_user", feature_names=["user"]),
        EmbeddingBagConfig(num_embeddings=50, embedding_dim=16, name="t_item", feature_names=["item"]),
    ]


class TestTwoTower:
    def test_train_step(self):
        model = TwoTowerTrain(
            TwoTower(
                EmbeddingBagCollection(tables=_tables()),
                query_features=["user"],
                candidate_features=["item"],
                layer_sizes=[32, 8],
            )
        )
        opt = torch.optim.Adam(model.parameters(), lr=1e-3)
        for _ in range(2):
            loss = model(_kjt())
            opt.zero_grad()
            loss.backward()
            opt.step()
        assert torch.isfinite(loss)

    def test_quantized_inference(self):
        tt = TwoTower(
            EmbeddingBagCollection(tables=_tables()),
            query_features=["user"],
            candidate_features=["item"],
            layer_sizes=[32, 8],
        )
        q_f, c_f = tt(_kjt())
        quantize_inference_model(tt)
        assert isinstance(tt.ebc, QuantEBC)
        q_q, c_q = tt(_kjt())
        torch.testing.assert_close(q_q, q_f, atol=5e-2, rtol=0.2)

    def test_sequence_two_tower(self):
        ec = EmbeddingCollection(
            tables=[
                EmbeddingConfig(num_embeddings=60, embedding_dim=16, name="t_hist", feature_names=["hist"])
            ]
        )
        model = SequenceTwoTower(
            ec,
            history_feature="hist",
            candidate_ebc=EmbeddingBagCollection(
                tables=[
                    EmbeddingBagConfig(num_embeddings=50, embedding_dim=16, name="t_item", feature_names=["item"])
                ]
            ),
            candidate_features=["item"],
            layer_sizes=[32, 8],
        )
        B = 4
        torch.manual_seed(1)
        hist_lengths = torch.randint(1, 6, (B,))
        history = KeyedJaggedTensor(
            keys=["hist"],
            values=torch.randint(0, 60, (int(hist_lengths.sum()),)),
            lengths=hist_lengths,
            stride=B,
        )
        cands = KeyedJaggedTensor(
            keys=["item"],
            values=torch.randint(0, 50, (B,)),
            lengths=torch.ones(B, dtype=torch.int64),
            stride=B,
        )
        q, c = model(history, cands)
        assert q.shape == (B, 8) and c.shape == (B, 8)
        (q * c).sum().backward()
