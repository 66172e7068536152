"""gRPC Predictor service over the native batching runtime.

Reference parity: torchrec/inference/server.cpp:44 (PredictorServiceHandler)
+ inference_legacy/src/{BatchingQueue,GPUExecutor,ResultSplit}.cpp. The wire
protocol is the reference's predictor.proto (see predictor_proto.py); the
hot path — request coalescing, pinned staging, execution, result split —
runs in the C++ extension (csrc/batching_queue.cpp), with one HIP stream per
executor thread so H2D copies and compute from different batches overlap.
"""

from __future__ import annotations

import concurrent.futures
import threading
from typing import Callable, Dict, List, Optional

import numpy as np
import torch

import grpc

from torchrec_amd.inference.predictor_proto import (
    PREDICT_METHOD,
    FloatVec,
    PredictionRequest,
    PredictionResponse,
)


def _sparse_to_tensors(sf, batch_size: int):
    """SparseFeatures {num_features, lengths: int32 [T*B], values: int64} ->
    per-feature (values, lengths) tensors (reference Batching.cpp layout)."""
    T = sf.num_features
    lengths = torch.from_numpy(
        np.frombuffer(sf.lengths, dtype=np.int32).copy()
    ).to(torch.int64)
    values = torch.from_numpy(np.frombuffer(sf.values, dtype=np.int64).copy())
    lengths = lengths.view(T, batch_size)
    offsets = torch.zeros(T * batch_size + 1, dtype=torch.int64)
    torch.cumsum(lengths.reshape(-1), 0, out=offsets[1:])
    per_feature_values = []
    per_feature_lengths = []
    for t in range(T):
        lo = int(offsets[t * batch_size])
        hi = int(offsets[(t + 1) * batch_size])
        per_feature_values.append(values[lo:hi])
        per_feature_lengths.append(lengths[t])
    return per_feature_values, per_feature_lengths


class PredictorService:
    """Predict handler: decode proto -> enqueue -> await split result."""

    def __init__(self, queue, task_names: List[str], num_dense: int) -> None:
        self._queue = queue
        self._tasks = task_names
        self._num_dense = num_dense

    def predict(self, request: PredictionRequest) -> PredictionResponse:
        B = request.batch_size
        dense = torch.from_numpy(
            np.frombuffer(request.float_features.values, dtype=np.float32).copy()
        ).view(B, self._num_dense)
        values, lengths = _sparse_to_tensors(request.id_list_features, B)
        fut = self._queue.add(dense, values, lengths)
        result = fut.get()  # GIL released inside (TensorFuture.get)
        resp = PredictionResponse()
        out = result.detach().float().cpu()
        if out.dim() == 1:
            out = out.unsqueeze(1)
        for i, task in enumerate(self._tasks):
            col = out[:, i] if out.shape[1] > i else out[:, 0]
            resp.predictions[task].data.extend(col.tolist())
        return resp


class PredictorServer:
    """grpc.Server hosting /predictor.Predictor/Predict with the runtime-built
    proto messages as (de)serializers."""

    def __init__(
        self,
        model: Callable[[torch.Tensor, torch.Tensor, torch.Tensor], torch.Tensor],
        num_features: int,
        num_dense: int,
        task_names: Optional[List[str]] = None,
        max_batch_size: int = 1024,
        batching_interval_ms: int = 2,
        num_exec_threads: int = 2,
        address: str = "127.0.0.1:0",
        device: Optional[torch.device] = None,
    ) -> None:
        from torchrec_amd.inference import _batching

        self._queue = _batching.BatchingQueue(
            num_features, max_batch_size, batching_interval_ms
        )
        dev = device or (
            torch.device("cuda") if torch.cuda.is_available() else torch.device("cpu")
        )
        streams = (
            [torch.cuda.Stream(device=dev) for _ in range(num_exec_threads)]
            if dev.type == "cuda"
            else [None] * num_exec_threads
        )
        tls = threading.local()

        def run_model(dense, values, lengths):
            # one HIP stream per executor thread: H2D of batch k+1 overlaps
            # compute of batch k (reference GPUExecutor.h:38 per-GPU threads)
            if dev.type == "cuda":
                if not hasattr(tls, "stream"):
                    with self._stream_lock:
                        tls.stream = streams[self._next_stream % len(streams)]
                        self._next_stream += 1
                with torch.cuda.stream(tls.stream):
                    out = model(
                        dense.to(dev, non_blocking=True),
                        values.to(dev, non_blocking=True),
                        lengths.to(dev, non_blocking=True),
                    )
                    tls.stream.synchronize()
                    return out
            return model(dense, values, lengths)

        self._stream_lock = threading.Lock()
        self._next_stream = 0
        self._executor = _batching.GPUExecutor(
            self._queue, run_model, num_exec_threads
        )
        self._service = PredictorService(
            self._queue, task_names or ["default"], num_dense
        )
        self._server = grpc.server(
            concurrent.futures.ThreadPoolExecutor(max_workers=16)
        )
        rpc = grpc.unary_unary_rpc_method_handler(
            lambda req, ctx: self._service.predict(req),
            request_deserializer=PredictionRequest.FromString,
            response_serializer=PredictionResponse.SerializeToString,
        )
        handler = grpc.method_handlers_generic_handler(
            "predictor.Predictor", {"Predict": rpc}
        )
        self._server.add_generic_rpc_handlers((handler,))
        self.port = self._server.add_insecure_port(address)
        self._server.start()

    def stop(self, grace: float = 1.0) -> None:
        self._server.stop(grace).wait()
        self._executor.join()


class PredictorClient:
    """Minimal client speaking the reference predictor.proto wire."""

    def __init__(self, address: str) -> None:
        self._channel = grpc.insecure_channel(address)
        self._predict = self._channel.unary_unary(
            PREDICT_METHOD,
            request_serializer=PredictionRequest.SerializeToString,
            response_deserializer=PredictionResponse.FromString,
        )

    def predict(
        self,
        dense: torch.Tensor,  # [B, num_dense] fp32
        sparse_lengths: torch.Tensor,  # [T, B] int
        sparse_values: torch.Tensor,  # flat int64, feature-major
    ) -> Dict[str, List[float]]:
        req = PredictionRequest(batch_size=dense.shape[0])
        req.float_features.num_features = dense.shape[1]
        req.float_features.values = dense.float().numpy().tobytes()
        req.id_list_features.num_features = sparse_lengths.shape[0]
        req.id_list_features.lengths = (
            sparse_lengths.to(torch.int32).numpy().tobytes()
        )
        req.id_list_features.values = sparse_values.to(torch.int64).numpy().tobytes()
        resp = self._predict(req)
        return {k: list(v.data) for k, v in resp.predictions.items()}

    def close(self) -> None:
        self._channel.close()
