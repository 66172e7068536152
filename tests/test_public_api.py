"""Public API surface guard: the imports a reference user relies on."""


def test_top_level_surface():
    import torchrec_amd as trec

    for name in (
        "JaggedTensor", "KeyedJaggedTensor", "KeyedTensor",
        "EmbeddingBagConfig", "EmbeddingConfig", "DataType", "PoolingType",
        "EmbeddingBagCollection", "EmbeddingCollection",
    ):
        assert hasattr(trec, name), name


def test_distributed_surface():
    import torchrec_amd.distributed as trec_dist

    for name in (
        "DistributedModelParallel", "DMPCollection", "ShardingEnv",
        "ShardingType", "ShardingPlan", "ParameterSharding", "ModuleSharder",
        "EmbeddingBagCollectionSharder", "EmbeddingCollectionSharder",
        "TrainPipelineSparseDist", "EvalPipelineSparseDist",
        "CacheParams", "KeyValueParams", "Awaitable", "LazyAwaitable",
    ):
        assert hasattr(trec_dist, name), name


def test_subpackage_surfaces():
    import torchrec_amd.metrics as m
    import torchrec_amd.modules as mod
    import torchrec_amd.optim as o
    import torchrec_amd.quant as q
    import torchrec_amd.inference as inf
    import torchrec_amd.datasets as d
    import torchrec_amd.ir as ir

    assert m.NEMetric and m.AUCMetric and m.RecMetricModule
    assert mod.FusedEmbeddingBagCollection and mod.MLP
    assert o.KeyedOptimizer and o.CombinedOptimizer and o.FusedOptimizer
    assert q.EmbeddingBagCollection and q.EmbeddingCollection
    assert inf.quantize_inference_model and inf.shard_quant_model
    assert d.RandomRecDataset and d.generate_batch
    assert ir.encapsulate_ir_modules and ir.decapsulate_ir_modules


def test_ops_surface():
    from torchrec_amd import ops

    for name in (
        "complete_cumsum", "permute_2d_sparse_data", "permute_1d_sparse_data",
        "jagged_to_padded_dense", "dense_to_jagged", "segment_sum_csr",
        "block_bucketize_sparse_features", "permute_pooled_embs",
        "fused_interaction", "jagged_index_select_2d", "jagged_unique_indices",
        "keyed_jagged_index_select_dim1", "group_index_select_dim0",
        "batch_index_select_dim0", "expand_into_jagged_permute",
        "invert_permute", "lengths_range",
    ):
        assert hasattr(ops, name), name
