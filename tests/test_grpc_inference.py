"""gRPC Predictor service tests (reference server.cpp:44 + predictor.proto
wire; C++ batching runtime underneath)."""

import threading

import pytest
import torch

grpc = pytest.importorskip("grpc")


def _toy_model(num_features=2, num_dense=4, dim=8):
    torch.manual_seed(0)
    emb = torch.randn(100, dim)
    w = torch.randn(num_dense + num_features * dim, 1)

    def model(dense, values, lengths):
        B = dense.shape[0]
        lengths2 = lengths.view(num_features, B)
        offsets = torch.zeros(num_features * B + 1, dtype=torch.int64)
        torch.cumsum(lengths2.reshape(-1), 0, out=offsets[1:])
        pooled = []
        for f in range(num_features):
            for b in range(B):
                lo, hi = int(offsets[f * B + b]), int(offsets[f * B + b + 1])
                rows = emb[values[lo:hi]]
                pooled.append(rows.sum(0) if hi > lo else torch.zeros(dim))
        pooled_t = torch.stack(pooled).view(num_features, B, dim).permute(1, 0, 2)
        x = torch.cat([dense, pooled_t.reshape(B, -1)], dim=1)
        return x @ w

    return model


class TestGrpcPredictor:
    def _request(self, B=3, T=2, num_dense=4, seed=0):
        g = torch.Generator().manual_seed(seed)
        dense = torch.randn(B, num_dense, generator=g)
        lengths = torch.randint(0, 3, (T, B), generator=g)
        values = torch.randint(0, 100, (int(lengths.sum()),), generator=g)
        return dense, lengths, values

    def test_predict_roundtrip_matches_local(self):
        from torchrec_amd.inference.grpc_server import (
            PredictorClient,
            PredictorServer,
        )

        model = _toy_model()
        server = PredictorServer(
            model, num_features=2, num_dense=4, task_names=["ctr"],
            num_exec_threads=2, device=torch.device("cpu"),
        )
        try:
            client = PredictorClient(f"127.0.0.1:{server.port}")
            dense, lengths, values = self._request(seed=1)
            out = client.predict(dense, lengths, values)
            assert "ctr" in out and len(out["ctr"]) == 3
            ref = model(dense, values, lengths.reshape(-1))
            torch.testing.assert_close(
                torch.tensor(out["ctr"]), ref.squeeze(1), atol=1e-5, rtol=1e-5
            )
            client.close()
        finally:
            server.stop()

    def test_concurrent_requests_batch_and_split(self):
        from torchrec_amd.inference.grpc_server import (
            PredictorClient,
            PredictorServer,
        )

        model = _toy_model()
        server = PredictorServer(
            model, num_features=2, num_dense=4, task_names=["ctr"],
            batching_interval_ms=5, num_exec_threads=2,
            device=torch.device("cpu"),
        )
        results = {}
        try:
            def call(i):
                c = PredictorClient(f"127.0.0.1:{server.port}")
                d, l, v = self._request(seed=10 + i)
                results[i] = (c.predict(d, l, v), model(d, v, l.reshape(-1)))
                c.close()

            threads = [threading.Thread(target=call, args=(i,)) for i in range(6)]
            for t in threads:
                t.start()
            for t in threads:
                t.join()
            assert len(results) == 6
            for i, (got, ref) in results.items():
                torch.testing.assert_close(
                    torch.tensor(got["ctr"]), ref.squeeze(1), atol=1e-5, rtol=1e-5
                )
        finally:
            server.stop()

    def test_wire_is_reference_proto_shape(self):
        from torchrec_amd.inference.predictor_proto import (
            PredictionRequest, PredictionResponse,
        )

        # field numbers match reference predictor.proto
        req = PredictionRequest()
        fields = {f.name: f.number for f in req.DESCRIPTOR.fields}
        assert fields == {
            "batch_size": 1, "float_features": 2, "id_list_features": 3,
            "id_score_list_features": 4, "embedding_features": 5,
            "unary_features": 6,
        }
        resp = PredictionResponse()
        assert [f.name for f in resp.DESCRIPTOR.fields] == ["predictions"]
