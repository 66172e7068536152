"""Native C++ batching runtime tests (reference:
inference_legacy/tests/BatchingQueueTest.cpp pattern, driven from Python)."""

import torch

from torchrec_amd.inference._batching import BatchingQueue, GPUExecutor


def test_batching_and_execution():
    F = 2
    q = BatchingQueue(num_features=F, max_batch_size=64, batching_interval_ms=1)

    def model(dense, values, lengths):
        # returns per-item sum of dense + count of sparse ids per item
        B = dense.shape[0]
        lengths2d = lengths.view(F, B)
        return dense.sum(dim=1, keepdim=True) + lengths2d.sum(dim=0, keepdim=True).t().float()

    ex = GPUExecutor(q, model, num_threads=1)
    futs = []
    expected = []
    for i in range(5):
        b = i % 2 + 1
        dense = torch.full((b, 3), float(i))
        values = [torch.arange(b * 2), torch.arange(b)]
        lengths = [torch.full((b,), 2, dtype=torch.int64), torch.ones(b, dtype=torch.int64)]
        futs.append(q.add(dense, values, lengths))
        expected.append(dense.sum(dim=1, keepdim=True) + 3.0)
    for fut, exp in zip(futs, expected):
        out = fut.get()
        torch.testing.assert_close(out, exp)
    ex.join()


def test_predictor_server():
    """End-to-end serve path: quantized model behind the HTTP Predict endpoint
    + C++ batching queue (reference server.cpp Predict semantics)."""
    from fastapi.testclient import TestClient

    from torchrec_amd.inference.modules import quantize_inference_model
    from torchrec_amd.inference.server import make_predictor_app
    from torchrec_amd.models.dlrm import DLRM
    from torchrec_amd.modules.embedding_configs import EmbeddingBagConfig
    from torchrec_amd.modules.embedding_modules import EmbeddingBagCollection
    from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor

    keys = ["f0", "f1"]
    tables = [
        EmbeddingBagConfig(num_embeddings=50, embedding_dim=8, name=f"t{i}", feature_names=[k])
        for i, k in enumerate(keys)
    ]
    model = DLRM(
        embedding_bag_collection=EmbeddingBagCollection(tables=tables),
        dense_in_features=4,
        dense_arch_layer_sizes=[8, 8],
        over_arch_layer_sizes=[8, 1],
    )
    quantize_inference_model(model)
    model.eval()

    F = len(keys)

    def model_fn(dense, values, lengths):
        B = dense.shape[0]
        kjt = KeyedJaggedTensor(keys=keys, values=values, lengths=lengths, stride=B)
        with torch.no_grad():
            return model(dense, kjt)

    app = make_predictor_app(model_fn, keys)
    with TestClient(app) as client:
        assert client.get("/health").json()["status"] == "ok"
        req = {
            "float_features": [[0.1, 0.2, 0.3, 0.4], [0.5, 0.6, 0.7, 0.8]],
            "id_list_features": {
                "f0": {"values": [1, 2, 3], "lengths": [2, 1]},
                "f1": {"values": [7], "lengths": [0, 1]},
            },
        }
        resp = client.post("/predict", json=req)
        assert resp.status_code == 200
        preds = resp.json()["predictions"]
        assert len(preds) == 2
        # matches direct model call
        kjt = KeyedJaggedTensor(
            keys=keys,
            values=torch.tensor([1, 2, 3, 7]),
            lengths=torch.tensor([2, 1, 0, 1]),
            stride=2,
        )
        with torch.no_grad():
            direct = model(torch.tensor(req["float_features"]), kjt).reshape(-1)
        torch.testing.assert_close(torch.tensor(preds), direct, atol=1e-5, rtol=1e-5)


def test_predict_factory_package_roundtrip(tmp_path):
    """torch.package export/import of a predict factory + configs."""
    from torchrec_amd.inference.model_packager import (
        PredictFactoryPackager,
        load_predict_factory,
    )

    class P(PredictFactoryPackager):
        @classmethod
        def set_extern_modules(cls):
            return []

        @classmethod
        def set_mocked_modules(cls):
            return []

    configs = {"tables": 4, "dim": 16}
    out = str(tmp_path / "factory.pt")
    # a pickleable payload standing in for a factory (classes defined in
    # tests aren't packageable; serving code passes an importable factory)
    P.save_predict_factory({"entry": "make_predictor", "version": 1}, configs, out)
    factory, cfg = load_predict_factory(out)
    assert factory["entry"] == "make_predictor"
    assert cfg == configs


def test_invoke_on_rank_and_broadcast():
    from tests.dist_utils import run_multi_process

    run_multi_process(_run_invoke_broadcast, 2, "gloo")


def _run_invoke_broadcast(rank, world_size):
    import torch.distributed as dist

    from torchrec_amd.distributed.collective_utils import (
        invoke_on_rank_and_broadcast_result,
    )

    result = invoke_on_rank_and_broadcast_result(
        dist.group.WORLD, 0, lambda: {"from": dist.get_rank(), "x": 42}
    )
    assert result == {"from": 0, "x": 42}  # every rank sees rank-0's value
