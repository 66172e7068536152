"""Cross-stream / pipeline contracts (reference: torchrec/streamable.py:15,30)."""

import abc

import torch


class Multistreamable(abc.ABC):
    """Objects whose tensors may be used across HIP streams.

    ``record_stream`` must forward to every contained tensor so the caching
    allocator does not reuse their memory while another stream still reads it.
    """

    @abc.abstractmethod
    def record_stream(self, stream: torch.Stream) -> None:
        ...


class Pipelineable(Multistreamable):
    """Input batches that a train pipeline can move across devices/streams."""

    @abc.abstractmethod
    def to(self, device: torch.device, non_blocking: bool) -> "Pipelineable":
        ...
