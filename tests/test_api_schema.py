"""Frozen-API-signature tests (reference torchrec/schema/api_tests/*.py —
7 files; same intent here: the stable parameter lists below are the contract
a reference user relies on, and changing them must break CI)."""

import inspect

import pytest


def _params(fn):
    return [
        p.name
        for p in inspect.signature(fn).parameters.values()
        if p.name not in ("self", "args", "kwargs")
    ]


def assert_sig_prefix(fn, stable):
    """Every stable name must appear in order in the actual signature."""
    actual = _params(fn)
    it = iter(actual)
    for name in stable:
        for got in it:
            if got == name:
                break
        else:
            raise AssertionError(
                f"{fn.__qualname__}: stable param {name!r} missing/out of order; "
                f"actual={actual}"
            )


class TestJaggedTensorSchema:
    def test_jagged_tensor(self):
        from torchrec_amd.sparse.jagged_tensor import JaggedTensor

        assert_sig_prefix(JaggedTensor.__init__, ["values", "weights", "lengths", "offsets"])
        assert_sig_prefix(JaggedTensor.to_padded_dense, ["desired_length", "padding_value"])

    def test_kjt(self):
        from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor

        assert_sig_prefix(
            KeyedJaggedTensor.__init__,
            ["keys", "values", "weights", "lengths", "offsets"],
        )
        assert_sig_prefix(KeyedJaggedTensor.split, ["segments"])
        assert_sig_prefix(KeyedJaggedTensor.permute, ["indices"])
        assert_sig_prefix(KeyedJaggedTensor.from_lengths_sync, ["keys", "values"])

    def test_kt(self):
        from torchrec_amd.sparse.jagged_tensor import KeyedTensor

        assert_sig_prefix(
            KeyedTensor.__init__, ["keys", "length_per_key", "values", "key_dim"]
        )
        assert_sig_prefix(KeyedTensor.regroup, ["keyed_tensors", "groups"])


class TestEmbeddingConfigSchema:
    def test_bag_config(self):
        from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig

        fields = list(EmbeddingBagConfig.__dataclass_fields__)
        for f in ("num_embeddings", "embedding_dim", "name", "data_type",
                  "feature_names", "pooling", "use_virtual_table",
                  "virtual_table_eviction_policy"):
            assert f in fields, f

    def test_config(self):
        from torchrec_amd.modules.embedding_configs import EmbeddingConfig

        for f in ("num_embeddings", "embedding_dim", "name", "feature_names"):
            assert f in EmbeddingConfig.__dataclass_fields__


class TestEmbeddingModuleSchema:
    def test_ebc(self):
        from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection

        assert_sig_prefix(EmbeddingBagCollection.__init__, ["tables", "is_weighted", "device"])
        assert_sig_prefix(EmbeddingBagCollection.forward, ["features"])

    def test_ec(self):
        from torchrec_amd.modules.embedding_modules import EmbeddingCollection

        assert_sig_prefix(EmbeddingCollection.__init__, ["tables"])
        assert_sig_prefix(EmbeddingCollection.forward, ["features"])


class TestModelParallelSchema:
    def test_dmp(self):
        from torchrec_amd.distributed.model_parallel import DistributedModelParallel

        assert_sig_prefix(
            DistributedModelParallel.__init__,
            ["module", "env", "device", "plan", "sharders"],
        )

    def test_dmp_collection(self):
        from torchrec_amd.distributed.model_parallel import DMPCollection

        assert_sig_prefix(
            DMPCollection.__init__,
            ["module", "sharding_group_size", "device", "plan", "sharders"],
        )

    def test_sharding_env(self):
        from torchrec_amd.distributed.types import ShardingEnv

        assert hasattr(ShardingEnv, "from_process_group")
        assert hasattr(ShardingEnv, "from_local")


class TestOptimizerSchema:
    def test_keyed(self):
        from torchrec_amd.optim.keyed import CombinedOptimizer, KeyedOptimizer

        assert_sig_prefix(KeyedOptimizer.__init__, ["params", "state", "param_groups"])
        for m in ("state_dict", "load_state_dict", "save_param_groups", "step",
                  "zero_grad", "init_state"):
            assert hasattr(KeyedOptimizer, m), m
        assert hasattr(CombinedOptimizer, "prepend_opt_key")

    def test_wrappers(self):
        from torchrec_amd.optim.optimizers import (
            GradientClippingOptimizer,
            RowWiseAdagrad,
            WarmupOptimizer,
        )

        assert_sig_prefix(
            GradientClippingOptimizer.__init__,
            ["optimizer", "clipping", "max_gradient", "norm_type"],
        )
        assert_sig_prefix(WarmupOptimizer.__init__, ["optimizer", "stages", "lr"])
        assert_sig_prefix(RowWiseAdagrad.__init__, ["params", "lr", "eps"])


class TestPlannerSchema:
    def test_planner(self):
        from torchrec_amd.distributed.planner.planners import EmbeddingShardingPlanner

        assert_sig_prefix(EmbeddingShardingPlanner.__init__, ["topology"])
        assert_sig_prefix(EmbeddingShardingPlanner.plan, ["module", "sharders"])
        assert_sig_prefix(
            EmbeddingShardingPlanner.collective_plan, ["module", "sharders", "pg"]
        )

    def test_topology(self):
        from torchrec_amd.distributed.planner.types import (
            ParameterConstraints,
            Topology,
        )

        assert_sig_prefix(Topology.__init__, ["world_size", "compute_device"])
        assert "sharding_types" in ParameterConstraints.__dataclass_fields__


class TestInferenceSchema:
    def test_modules(self):
        from torchrec_amd.inference.modules import (
            PredictFactory,
            PredictModule,
            quantize_inference_model,
            shard_quant_model,
        )

        assert_sig_prefix(quantize_inference_model, ["model"])
        assert_sig_prefix(shard_quant_model, ["model"])
        assert hasattr(PredictModule, "predict_forward")
        assert hasattr(PredictFactory, "create_predict_module")

    def test_quant_modules(self):
        from torchrec_amd.quant.embedding_modules import EmbeddingBagCollection

        assert hasattr(EmbeddingBagCollection, "from_float")


class TestCheckpointSchema:
    def test_dcp_helpers(self):
        from torchrec_amd.distributed.checkpoint import (
            load_checkpoint,
            save_checkpoint,
            state_dict_for_checkpoint,
        )

        assert_sig_prefix(save_checkpoint, ["model", "path"])
        assert_sig_prefix(load_checkpoint, ["model", "path"])
        assert_sig_prefix(state_dict_for_checkpoint, ["model"])


class TestPsTransportSchema:
    def test_ps_io(self):
        from torchrec_amd.dynamic_embedding.ps import (
            ParameterServer,
            PSIO,
            get_ps_io,
            register_ps_io,
        )
        from torchrec_amd.dynamic_embedding.ps_net import PSNetServer, TcpPSIO

        assert_sig_prefix(ParameterServer.__init__, ["dims", "io"])
        assert_sig_prefix(get_ps_io, ["name", "dim"])
        assert_sig_prefix(TcpPSIO.__init__, ["dim", "address"])
        assert hasattr(PSNetServer, "address") and hasattr(PSNetServer, "close")
        assert hasattr(PSIO, "push") and hasattr(PSIO, "pull")


class TestStaticSplitsSchema:
    def test_dist_data(self):
        from torchrec_amd.distributed.dist_data import (
            KJTAllToAll,
            set_static_kjt_splits,
            static_kjt_splits_enabled,
        )

        assert_sig_prefix(KJTAllToAll.__init__, ["pg", "splits"])
        assert "allow_static" in [
            p.name
            for p in __import__("inspect").signature(KJTAllToAll.__init__).parameters.values()
        ]
        set_static_kjt_splits(False)
        assert static_kjt_splits_enabled() in (True, False)
