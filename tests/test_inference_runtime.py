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
