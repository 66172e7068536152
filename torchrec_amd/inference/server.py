"""Inference server: Predict endpoint over the native batching runtime.

Reference parity: torchrec/inference/server.cpp:44 (PredictorServiceHandler —
a Predict RPC wrapping the model) + inference/client.py. grpcio-tools is not
available in this image, so the wire is HTTP/JSON via FastAPI with the same
request/response shape as the reference's predictor.proto:
  request:  {"float_features": [[...]], "id_list_features": {feat: {"values":
             [...], "lengths": [...]}}}
  response: {"predictions": [...]}
Requests are funneled through the C++ BatchingQueue + GPUExecutor
(torchrec_amd/inference/csrc/batching_queue.cpp) so concurrent callers are
cross-request batched before hitting the GPU.
"""

from typing import Callable, Dict, List, Optional

import torch


def make_predictor_app(
    model_fn: Callable[[torch.Tensor, torch.Tensor, torch.Tensor], torch.Tensor],
    feature_names: List[str],
    max_batch_size: int = 2048,
    batching_interval_ms: int = 2,
    num_exec_threads: int = 1,
):
    """Build the FastAPI app around a (dense, values, lengths) -> predictions
    callable (typically a quantized sharded model)."""
    from fastapi import FastAPI, Request

    from torchrec_amd.inference._batching import BatchingQueue, GPUExecutor

    queue = BatchingQueue(
        num_features=len(feature_names),
        max_batch_size=max_batch_size,
        batching_interval_ms=batching_interval_ms,
    )
    executor = GPUExecutor(queue, model_fn, num_threads=num_exec_threads)

    app = FastAPI(title="torchrec_amd predictor")
    app.state.queue = queue
    app.state.executor = executor

    @app.post("/predict")
    async def predict(request: Request) -> Dict[str, List[float]]:
        req = await request.json()
        dense = torch.tensor(req["float_features"], dtype=torch.float32)
        values = [
            torch.tensor(req["id_list_features"][f]["values"], dtype=torch.int64)
            for f in feature_names
        ]
        lengths = [
            torch.tensor(req["id_list_features"][f]["lengths"], dtype=torch.int64)
            for f in feature_names
        ]
        fut = queue.add(dense, values, lengths)
        out = fut.get()
        return {"predictions": out.reshape(-1).tolist()}

    @app.get("/health")
    def health() -> Dict[str, str]:
        return {"status": "ok"}

    @app.on_event("shutdown")
    def shutdown() -> None:
        executor.join()

    return app


def serve(app, host: str = "127.0.0.1", port: int = 50051) -> None:
    import uvicorn

    uvicorn.run(app, host=host, port=port)
