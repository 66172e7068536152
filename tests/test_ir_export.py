"""torch.export-integrated IR round trip (reference: torchrec/ir/tests/
test_serializer.py pattern — encapsulate -> export -> unflatten ->
decapsulate -> numerical parity)."""

import torch
import torch.nn as nn

from torchrec_amd.ir.utils import (
    decapsulate_ir_modules,
    encapsulate_ir_modules,
    mark_dynamic_kjt,
    register_kjt_pytree,
)
from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig
from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection
from torchrec_amd.modules.feature_processor import (
    FeatureProcessedEmbeddingBagCollection,
    PositionWeightedModuleCollection,
)
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor


def make_tables(weighted=False):
    return [
        EmbeddingBagConfig(num_embeddings=20, embedding_dim=8, name="t0",
                           feature_names=["f0"]),
        EmbeddingBagConfig(num_embeddings=30, embedding_dim=12, name="t1",
                           feature_names=["f1"]),
    ]


def make_kjt(B=4, seed=0):
    g = torch.Generator().manual_seed(seed)
    lengths = torch.randint(1, 4, (2 * B,), generator=g)
    values = torch.randint(0, 20, (int(lengths.sum()),), generator=g)
    return KeyedJaggedTensor(keys=["f0", "f1"], values=values, lengths=lengths, stride=B)


class Wrapper(nn.Module):
    def __init__(self, sparse):
        super().__init__()
        self.sparse = sparse
        self.dense = nn.Linear(20, 4)

    def forward(self, kjt):
        kt = self.sparse(kjt)
        return self.dense(kt.values())


class TestIrExport:
    def _roundtrip(self, sparse_module):
        model = Wrapper(sparse_module)
        ref_out = model(make_kjt(B=4, seed=1)).detach()
        ref_state = {k: v.clone() for k, v in model.state_dict().items()
                     if "ir_metadata" not in k}

        model, preserved = encapsulate_ir_modules(model)
        assert preserved, "no embedding modules encapsulated"
        kjt = make_kjt(B=4, seed=0)
        shapes = mark_dynamic_kjt(kjt, variable_batch=True)
        ep = torch.export.export(
            model, (kjt,), dynamic_shapes=shapes.dynamic_shapes(model, (kjt,)),
            strict=False, preserve_module_call_signature=tuple(preserved),
        )
        # exported graph must contain the meta-stub op, not real embedding ops
        ops = {str(n.target) for n in ep.graph.nodes if n.op == "call_function"}
        assert any("ir_emb_lookup" in o for o in ops), ops

        unflat = torch.export.unflatten(ep)
        rebuilt = decapsulate_ir_modules(unflat)
        rebuilt.load_state_dict(ref_state, strict=False)
        out = rebuilt(make_kjt(B=4, seed=1))
        torch.testing.assert_close(out, ref_out, atol=1e-6, rtol=1e-6)
        # dynamic batch: a different values length must run through the graph
        out6 = rebuilt(make_kjt(B=6, seed=3))
        assert out6.shape[0] == 6

    def test_ebc_export_roundtrip(self):
        torch.manual_seed(0)
        self._roundtrip(EmbeddingBagCollection(tables=make_tables()))

    def test_fpebc_export_roundtrip(self):
        torch.manual_seed(0)
        ebc = EmbeddingBagCollection(tables=make_tables(), is_weighted=True)
        fp = PositionWeightedModuleCollection({"f0": 4, "f1": 4})
        self._roundtrip(FeatureProcessedEmbeddingBagCollection(ebc, fp))

    def test_meta_stub_shapes(self):
        register_kjt_pytree()
        ebc = EmbeddingBagCollection(tables=make_tables())
        encapsulate_ir_modules(ebc)
        kt = ebc(make_kjt(B=5))
        assert kt.values().shape == (5, 20)
        assert kt.keys() == ["f0", "f1"]
