"""EBC / EC / DLRM module tests (mirrors reference modules tests)."""

import torch

from torchrec_amd.datasets.random import generate_batch
from torchrec_amd.models.dlrm import DLRM, DLRM_DCN, DLRMTrain, InteractionArch
from torchrec_amd.modules.embedding_configs import (
    EmbeddingBagConfig,
    EmbeddingConfig,
    PoolingType,
)
from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection, EmbeddingCollection
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


def make_kjt():
    return KeyedJaggedTensor(
        keys=["f1", "f2"],
        values=torch.tensor([1, 2, 3, 4, 5, 6]),
        lengths=torch.tensor([2, 0, 1, 1, 2, 0]),
        stride=3,
    )


class TestEmbeddingBagCollection:
    def test_forward_sum(self):
        tables = [
            EmbeddingBagConfig(num_embeddings=10, embedding_dim=4, name="t1", feature_names=["f1"]),
            EmbeddingBagConfig(num_embeddings=10, embedding_dim=8, name="t2", feature_names=["f2"]),
        ]
        ebc = EmbeddingBagCollection(tables=tables)
        kt = ebc(make_kjt())
        assert kt.keys() == ["f1", "f2"]
        assert kt.values().shape == (3, 12)
        # numerics vs manual lookup
        w1 = ebc.embedding_bags["t1"].weight
        expected_row0 = w1[1] + w1[2]
        assert torch.allclose(kt["f1"][0], expected_row0)
        assert torch.allclose(kt["f1"][1], torch.zeros(4))

    def test_forward_mean(self):
        tables = [
            EmbeddingBagConfig(
                num_embeddings=10,
                embedding_dim=4,
                name="t1",
                feature_names=["f1"],
                pooling=PoolingType.MEAN,
            ),
        ]
        ebc = EmbeddingBagCollection(tables=tables)
        kjt = KeyedJaggedTensor(
            keys=["f1"], values=torch.tensor([1, 2]), lengths=torch.tensor([2]), stride=1
        )
        kt = ebc(kjt)
        w = ebc.embedding_bags["t1"].weight
        assert torch.allclose(kt["f1"][0], (w[1] + w[2]) / 2)

    def test_weighted(self):
        tables = [
            EmbeddingBagConfig(num_embeddings=10, embedding_dim=4, name="t1", feature_names=["f1"]),
        ]
        ebc = EmbeddingBagCollection(tables=tables, is_weighted=True)
        kjt = KeyedJaggedTensor(
            keys=["f1"],
            values=torch.tensor([1, 2]),
            weights=torch.tensor([2.0, 3.0]),
            lengths=torch.tensor([2]),
            stride=1,
        )
        kt = ebc(kjt)
        w = ebc.embedding_bags["t1"].weight
        assert torch.allclose(kt["f1"][0], 2.0 * w[1] + 3.0 * w[2])

    def test_shared_table_multiple_features(self):
        tables = [
            EmbeddingBagConfig(
                num_embeddings=10, embedding_dim=4, name="t1", feature_names=["f1", "f2"]
            ),
        ]
        ebc = EmbeddingBagCollection(tables=tables)
        kt = ebc(make_kjt())
        assert kt.keys() == ["f1", "f2"]
        assert kt.values().shape == (3, 8)


class TestEmbeddingCollection:
    def test_forward(self):
        tables = [
            EmbeddingConfig(num_embeddings=10, embedding_dim=4, name="t1", feature_names=["f1"]),
            EmbeddingConfig(num_embeddings=10, embedding_dim=4, name="t2", feature_names=["f2"]),
        ]
        ec = EmbeddingCollection(tables=tables)
        out = ec(make_kjt())
        assert set(out.keys()) == {"f1", "f2"}
        assert out["f1"].values().shape == (3, 4)
        assert out["f1"].lengths().tolist() == [2, 0, 1]
        w = ec.embeddings["t1"].weight
        assert torch.allclose(out["f1"].values()[0], w[1])


class TestDLRM:
    def _tables(self, dim=8):
        return [
            EmbeddingBagConfig(
                num_embeddings=100, embedding_dim=dim, name=f"t{i}", feature_names=[f"f{i}"]
            )
            for i in range(3)
        ]

    def test_forward_shape(self):
        ebc = EmbeddingBagCollection(tables=self._tables())
        model = DLRM(
            embedding_bag_collection=ebc,
            dense_in_features=13,
            dense_arch_layer_sizes=[16, 8],
            over_arch_layer_sizes=[16, 1],
        )
        batch = generate_batch(
            keys=["f0", "f1", "f2"], batch_size=4, hash_sizes=[100] * 3, ids_per_feature=5
        )
        logits = model(batch.dense_features, batch.sparse_features)
        assert logits.shape == (4, 1)

    def test_interaction_math(self):
        ia = InteractionArch(num_sparse_features=2)
        dense = torch.tensor([[1.0, 0.0]])
        sparse = torch.tensor([[[0.0, 1.0], [1.0, 1.0]]])
        out = ia(dense, sparse)
        # pairs: (dense,s0)=0, (dense,s1)=1, (s0,s1)=1
        assert out.shape == (1, 2 + 3)
        assert out[0, 2:].tolist() == [0.0, 1.0, 1.0]

    def test_train_step(self):
        ebc = EmbeddingBagCollection(tables=self._tables())
        model = DLRMTrain(
            DLRM(
                embedding_bag_collection=ebc,
                dense_in_features=13,
                dense_arch_layer_sizes=[16, 8],
                over_arch_layer_sizes=[16, 1],
            )
        )
        opt = torch.optim.SGD(model.parameters(), lr=0.1)
        batch = generate_batch(
            keys=["f0", "f1", "f2"], batch_size=4, hash_sizes=[100] * 3, ids_per_feature=5
        )
        loss, _ = model(batch)
        loss.backward()
        opt.step()
        assert torch.isfinite(loss)

    def test_dlrm_dcn(self):
        ebc = EmbeddingBagCollection(tables=self._tables())
        model = DLRM_DCN(
            embedding_bag_collection=ebc,
            dense_in_features=13,
            dense_arch_layer_sizes=[16, 8],
            over_arch_layer_sizes=[16, 1],
            dcn_num_layers=2,
            dcn_low_rank_dim=4,
        )
        batch = generate_batch(
            keys=["f0", "f1", "f2"], batch_size=4, hash_sizes=[100] * 3, ids_per_feature=5
        )
        logits = model(batch.dense_features, batch.sparse_features)
        assert logits.shape == (4, 1)

    def test_dlrm_projection(self):
        from torchrec_amd.models.dlrm import DLRM_Projection

        ebc = EmbeddingBagCollection(tables=self._tables())
        model = DLRM_Projection(
            embedding_bag_collection=ebc,
            dense_in_features=13,
            dense_arch_layer_sizes=[16, 8],
            over_arch_layer_sizes=[16, 1],
            interaction_branch1_layer_sizes=[24, 16],  # I1 = 16/8 = 2
            interaction_branch2_layer_sizes=[24, 24],  # I2 = 24/8 = 3
        )
        batch = generate_batch(
            keys=["f0", "f1", "f2"], batch_size=4, hash_sizes=[100] * 3, ids_per_feature=5
        )
        logits = model(batch.dense_features, batch.sparse_features)
        assert logits.shape == (4, 1)
        # over-arch input = D + I1*I2 = 8 + 6
        assert model.over_arch.model[0]._mlp[0]._linear.in_features == 14
        logits.sum().backward()
        assert all(
            p.grad is not None
            for p in model.inter_arch.parameters()
        )


class TestFeatureProcessors:
    def test_position_weighted_module(self):
        from torchrec_amd.modules.feature_processor import PositionWeightedModule
        from torchrec_amd.sparse.jagged_tensor import JaggedTensor

        pw = PositionWeightedModule(max_feature_length=4)
        with torch.no_grad():
            pw.position_weight.copy_(torch.tensor([1.0, 2.0, 3.0, 4.0]))
        jt = JaggedTensor(values=torch.tensor([10, 11, 12]), lengths=torch.tensor([2, 1]))
        out = pw(jt)
        assert out.weights().tolist() == [1.0, 2.0, 1.0]

    def test_fp_ebc(self):
        from torchrec_amd.modules.feature_processor import (
            FeatureProcessedEmbeddingBagCollection,
            PositionWeightedModuleCollection,
        )

        tables = [
            EmbeddingBagConfig(num_embeddings=10, embedding_dim=4, name="t1", feature_names=["f1"]),
        ]
        ebc = EmbeddingBagCollection(tables=tables, is_weighted=True)
        fp = PositionWeightedModuleCollection({"f1": 8})
        fp_ebc = FeatureProcessedEmbeddingBagCollection(ebc, fp)
        kjt = KeyedJaggedTensor(
            keys=["f1"], values=torch.tensor([1, 2]), lengths=torch.tensor([2]), stride=1
        )
        kt = fp_ebc(kjt)
        w = ebc.embedding_bags["t1"].weight
        # default position weights are all 1 -> plain sum
        assert torch.allclose(kt["f1"][0], w[1] + w[2])
        # gradient flows to position weights
        kt.values().sum().backward()
        assert fp.position_weights["f1"].grad is not None


class TestDeepFMAndCross:
    def test_simple_deepfm(self):
        from torchrec_amd.modules.deepfm import SimpleDeepFMNN

        tables = [
            EmbeddingBagConfig(num_embeddings=50, embedding_dim=8, name=f"t{i}", feature_names=[f"f{i}"])
            for i in range(2)
        ]
        model = SimpleDeepFMNN(
            num_dense_features=4,
            embedding_bag_collection=EmbeddingBagCollection(tables=tables),
            hidden_layer_size=16,
            deep_fm_dimension=8,
        )
        kjt = KeyedJaggedTensor(
            keys=["f0", "f1"],
            values=torch.tensor([1, 2, 3, 4]),
            lengths=torch.tensor([1, 1, 1, 1]),
            stride=2,
        )
        out = model(torch.rand(2, 4), kjt)
        assert out.shape == (2, 1)
        out.sum().backward()

    def test_crossnets(self):
        from torchrec_amd.modules.crossnet import CrossNet, VectorCrossNet

        x = torch.randn(4, 8)
        assert CrossNet(8, 2)(x).shape == (4, 8)
        assert VectorCrossNet(8, 2)(x).shape == (4, 8)

    def test_fx_tracer(self):
        from torchrec_amd.fx.tracer import symbolic_trace

        tables = [
            EmbeddingBagConfig(num_embeddings=10, embedding_dim=4, name="t", feature_names=["f"])
        ]

        class M(torch.nn.Module):
            def __init__(self):
                super().__init__()
                self.ebc = EmbeddingBagCollection(tables=tables)
                self.lin = torch.nn.Linear(4, 1)

            def forward(self, kjt):
                return self.lin(self.ebc(kjt).values())

        gm = symbolic_trace(M())
        kjt = KeyedJaggedTensor(
            keys=["f"], values=torch.tensor([1, 2]), lengths=torch.tensor([2]), stride=1
        )
        out = gm(kjt)
        assert out.shape == (1, 1)


class TestVbeFusedEBC:
    def test_vbe_forward(self):
        from torchrec_amd.modules.fused_embedding_modules import FusedEmbeddingBagCollection

        torch.manual_seed(0)
        tables = [
            EmbeddingBagConfig(num_embeddings=30, embedding_dim=8, name="t0", feature_names=["f0"]),
            EmbeddingBagConfig(num_embeddings=40, embedding_dim=4, name="t1", feature_names=["f1"]),
        ]
        ebc = FusedEmbeddingBagCollection(tables, optimizer="sgd")
        # f0 batch 2, f1 batch 3 (variable stride)
        kjt = KeyedJaggedTensor(
            keys=["f0", "f1"],
            values=torch.tensor([1, 2, 3, 4, 5]),
            lengths=torch.tensor([2, 1, 1, 0, 1]),
            stride_per_key_per_rank=[[2], [3]],
        )
        kt = ebc(kjt)
        assert kt.key_dim() == 0
        assert kt.length_per_key() == [2 * 8, 3 * 4]
        w0 = ebc.split_embedding_weights()[0]
        torch.testing.assert_close(kt["f0"][:8], w0[1] + w0[2])
