"""Stream-overlapped train pipelines.

Reference parity: torchrec/distributed/train_pipeline/train_pipelines.py
(TrainPipelineBase :260 — 2-stage H2D overlap; TrainPipelineSparseDist :530 —
3-stage, 2 batches in flight: H2D on a memcpy stream ‖ sparse input_dist on a
data-dist stream ‖ fwd/bwd on the default stream, with sharded-module
forwards swapped for PipelinedForward consuming the pre-started input_dist).

MI355X notes: streams are HIP streams; the KJT a2a's splits phase syncs a
small tensor to host — it runs one batch ahead so the sync hides under the
current batch's compute. RCCL collectives ordered via work.wait() on the
consuming stream.
"""

from __future__ import annotations

import contextlib
import logging
from typing import Any, Callable, Dict, Iterator, List, Optional, Tuple

import torch

from torchrec_amd.distributed.types import LazyAwaitable
from torchrec_amd.sparse.jagged_tensor import KeyedJaggedTensor
from torchrec_amd.streamable import Pipelineable

logger = logging.getLogger(__name__)


def _wait_for_batch(batch: Pipelineable, stream: Optional[torch.cuda.Stream]) -> None:
    """Reference parity: train_pipeline/utils.py:151."""
    if stream is None:
        return
    torch.cuda.current_stream().wait_stream(stream)
    cur = torch.cuda.current_stream()
    batch.record_stream(cur)


class TrainPipelineBase:
    """2-stage: H2D copy of batch i+1 overlaps fwd/bwd of batch i
    (reference train_pipelines.py:260)."""

    def __init__(
        self,
        model: torch.nn.Module,
        optimizer: torch.optim.Optimizer,
        device: torch.device,
        autocast_dtype: Optional[torch.dtype] = None,
    ) -> None:
        self._model = model
        self._optimizer = optimizer
        self._device = device
        self._autocast_dtype = autocast_dtype
        self._memcpy_stream: Optional[torch.cuda.Stream] = (
            torch.cuda.Stream(priority=-1) if device.type == "cuda" else None
        )
        self._cur_batch: Optional[Pipelineable] = None
        self._connected = False

    def _autocast_ctx(self):
        if self._autocast_dtype is None:
            return contextlib.nullcontext()
        return torch.autocast(device_type=self._device.type, dtype=self._autocast_dtype)

    def _connect(self, dataloader_iter: Iterator[Pipelineable]) -> None:
        cur_batch = next(dataloader_iter)
        with torch.cuda.stream(self._memcpy_stream):
            self._cur_batch = cur_batch.to(self._device, non_blocking=True)
        self._connected = True

    def progress(self, dataloader_iter: Iterator[Pipelineable]) -> Any:
        if not self._connected:
            self._connect(dataloader_iter)
        cur_batch = self._cur_batch
        assert cur_batch is not None
        _wait_for_batch(cur_batch, self._memcpy_stream)

        with torch.autograd.profiler.record_function("## next-batch H2D ##"):
            try:
                next_batch = next(dataloader_iter)
            except StopIteration:
                next_batch = None
            if next_batch is not None:
                with torch.cuda.stream(self._memcpy_stream):
                    self._cur_batch = next_batch.to(self._device, non_blocking=True)
            else:
                self._cur_batch = None

        self._optimizer.zero_grad(set_to_none=True)
        with torch.autograd.profiler.record_function("## forward ##"), self._autocast_ctx():
            losses, output = self._model(cur_batch)
        with torch.autograd.profiler.record_function("## backward ##"):
            torch.sum(losses).backward()
        with torch.autograd.profiler.record_function("## optimizer ##"):
            self._optimizer.step()
        return output


class PipelinedForward:
    """Replaces a sharded module's forward: consumes the input_dist that the
    pipeline started one step ahead (reference runtime_forwards.py:106)."""

    def __init__(self, fqn: str, module: torch.nn.Module, pipeline: "TrainPipelineSparseDist") -> None:
        self._fqn = fqn
        self._module = module
        self._pipeline = pipeline

    def __call__(self, *args, **kwargs):
        ctx, tensors_awaitable = self._pipeline._fetch_dist(self._fqn)
        with torch.autograd.profiler.record_function(f"## wait_sparse_data_dist {self._fqn} ##"):
            dist_input = tensors_awaitable.wait()
        if self._pipeline._data_dist_stream is not None:
            torch.cuda.current_stream().wait_stream(self._pipeline._data_dist_stream)
            for kjt in dist_input if isinstance(dist_input, list) else [dist_input]:
                if hasattr(kjt, "record_stream"):
                    kjt.record_stream(torch.cuda.current_stream())
        return self._module.compute_and_output_dist(ctx, dist_input)


class TrainPipelineSparseDist:
    """3-stage pipeline, 2 batches in flight (reference train_pipelines.py:530).

    batch i:   fwd/bwd/opt on the default stream
    batch i+1: sparse input_dist (KJT a2a) on the data-dist stream
    batch i+2: H2D copy on the memcpy stream
    """

    def __init__(
        self,
        model: torch.nn.Module,
        optimizer: torch.optim.Optimizer,
        device: torch.device,
        execute_all_batches: bool = True,
        extract_kjt: Optional[Callable[[Any], KeyedJaggedTensor]] = None,
        autocast_dtype: Optional[torch.dtype] = None,
    ) -> None:
        self._model = model
        self._optimizer = optimizer
        self._device = device
        self._autocast_dtype = autocast_dtype
        self._execute_all_batches = execute_all_batches
        self._extract_kjt = extract_kjt or (lambda batch: batch.sparse_features)
        is_cuda = device.type == "cuda"
        self._memcpy_stream = torch.cuda.Stream(priority=-1) if is_cuda else None
        self._data_dist_stream = torch.cuda.Stream(priority=-1) if is_cuda else None
        self._batches: List[Optional[Pipelineable]] = []
        self._dist_contexts: List[Dict[str, Tuple[Any, Any]]] = []
        self._pipelined: Dict[str, torch.nn.Module] = {}
        self._rewritten = False

    # -- model rewrite -----------------------------------------------------

    def _rewrite_model(self) -> None:
        """Swap sharded-module forwards for PipelinedForward
        (reference utils.py:508 _rewrite_model)."""
        for fqn, module in self._model.named_modules():
            if hasattr(module, "compute_and_output_dist") and hasattr(module, "input_dist"):
                self._pipelined[fqn] = module
                module.forward = PipelinedForward(fqn, module, self)
        if not self._pipelined:
            logger.warning("TrainPipelineSparseDist: no sharded modules found")
        self._rewritten = True

    def _fetch_dist(self, fqn: str) -> Tuple[Any, Any]:
        return self._dist_contexts[0][fqn]

    # -- stages ------------------------------------------------------------

    def _copy_batch_to_gpu(self, dataloader_iter) -> Optional[Pipelineable]:
        with torch.autograd.profiler.record_function("## copy_batch_to_gpu ##"):
            try:
                batch = next(dataloader_iter)
            except StopIteration:
                return None
            with torch.cuda.stream(self._memcpy_stream):
                return batch.to(self._device, non_blocking=True)

    def _start_sparse_data_dist(self, batch: Optional[Pipelineable]) -> Dict[str, Tuple[Any, Any]]:
        """Reference parity: utils.py:248 _start_data_dist."""
        if batch is None:
            return {}
        out: Dict[str, Tuple[Any, Any]] = {}
        with torch.autograd.profiler.record_function("## start_sparse_data_dist ##"):
            with torch.cuda.stream(self._data_dist_stream):
                _wait_for_batch(batch, self._memcpy_stream)
                kjt = self._extract_kjt(batch)
                for fqn, module in self._pipelined.items():
                    ctx = module.create_context()
                    splits_aw = module.input_dist(ctx, kjt)
                    tensors_aw = splits_aw.wait()  # syncs small splits; one batch ahead
                    out[fqn] = (ctx, tensors_aw)
        return out

    def _fill_pipeline(self, dataloader_iter) -> None:
        b0 = self._copy_batch_to_gpu(dataloader_iter)
        if b0 is None:
            raise StopIteration
        if not self._rewritten:
            self._rewrite_model()
        d0 = self._start_sparse_data_dist(b0)
        b1 = self._copy_batch_to_gpu(dataloader_iter)
        self._batches = [b0, b1]
        self._dist_contexts = [d0]

    def progress(self, dataloader_iter: Iterator[Pipelineable]) -> Any:
        if not self._batches:
            self._fill_pipeline(dataloader_iter)
        cur_batch = self._batches[0]
        next_batch = self._batches[1]
        if cur_batch is None:
            raise StopIteration

        # stage 3 fill: batch i+2 H2D
        batch_ip2 = self._copy_batch_to_gpu(dataloader_iter)

        _wait_for_batch(cur_batch, self._memcpy_stream)
        self._optimizer.zero_grad(set_to_none=True)

        # batch i+1 sparse dist overlaps batch i compute
        d_next = self._start_sparse_data_dist(next_batch)

        ac = (
            torch.autocast(device_type=self._device.type, dtype=self._autocast_dtype)
            if self._autocast_dtype is not None
            else contextlib.nullcontext()
        )
        with torch.autograd.profiler.record_function("## forward ##"), ac:
            losses, output = self._model(cur_batch)
        with torch.autograd.profiler.record_function("## backward ##"):
            torch.sum(losses).backward()
        with torch.autograd.profiler.record_function("## optimizer ##"):
            self._optimizer.step()

        self._batches = [next_batch, batch_ip2]
        self._dist_contexts = [d_next]
        return output


class _StreamJoinedAwaitable(LazyAwaitable):
    """Joins the embedding-lookup stream into the consumer stream at wait."""

    def __init__(self, inner, stream) -> None:
        super().__init__()
        self._inner = inner
        self._stream = stream

    def _wait_impl(self):
        out = self._inner.wait()
        if self._stream is not None:
            torch.cuda.current_stream().wait_stream(self._stream)
            vals = out.values() if hasattr(out, "values") else out
            if isinstance(vals, torch.Tensor):
                vals.record_stream(torch.cuda.current_stream())
        return out


class FusedPipelinedForward(PipelinedForward):
    """compute_and_output_dist on a dedicated emb_lookup stream
    (reference TrainPipelineFusedSparseDist, train_pipelines.py:1424)."""

    def __call__(self, *args, **kwargs):
        ctx, tensors_awaitable = self._pipeline._fetch_dist(self._fqn)
        with torch.autograd.profiler.record_function(
            f"## wait_sparse_data_dist {self._fqn} ##"
        ):
            dist_input = tensors_awaitable.wait()
        ls = getattr(self._pipeline, "_emb_lookup_stream", None)
        if ls is None:
            if self._pipeline._data_dist_stream is not None:
                torch.cuda.current_stream().wait_stream(self._pipeline._data_dist_stream)
            return self._module.compute_and_output_dist(ctx, dist_input)
        ls.wait_stream(self._pipeline._data_dist_stream)
        with torch.cuda.stream(ls):
            for kjt in dist_input if isinstance(dist_input, list) else [dist_input]:
                if hasattr(kjt, "record_stream"):
                    kjt.record_stream(ls)
            with torch.autograd.profiler.record_function("## emb_lookup ##"):
                aw = self._module.compute_and_output_dist(ctx, dist_input)
        return _StreamJoinedAwaitable(aw, ls)


class TrainPipelineFusedSparseDist(TrainPipelineSparseDist):
    """SparseDist + a dedicated embedding-lookup stream: the TBE gather and
    pooled output a2a of batch i run concurrent with the dense layers that
    do not consume them yet (reference train_pipelines.py:1424)."""

    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self._emb_lookup_stream = (
            torch.cuda.Stream(priority=-1) if self._device.type == "cuda" else None
        )

    def _rewrite_model(self) -> None:
        for fqn, module in self._model.named_modules():
            if hasattr(module, "compute_and_output_dist") and hasattr(module, "input_dist"):
                self._pipelined[fqn] = module
                module.forward = FusedPipelinedForward(fqn, module, self)
        self._rewritten = True


class GradientAccumulationPipeline:
    """Gradient-accumulation wrapper (reference gradient_accumulation.py):
    dense optimizer steps every ``accumulation_steps`` batches with loss
    scaled by 1/N. NOTE (reference semantics): FUSED sparse optimizers apply
    their update inside every backward — accumulation only defers the dense
    step, exactly as in the reference wrapper."""

    def __init__(self, pipeline, accumulation_steps: int) -> None:
        assert accumulation_steps >= 1
        self._pipeline = pipeline
        self._n = accumulation_steps
        self._i = 0
        self._opt = pipeline._optimizer

        class _Ctl:
            """zero_grad is ALWAYS a no-op inside the pipeline (it would
            wipe the accumulation right before the boundary backward);
            step fires only on window boundaries."""

            def __init__(self, opt):
                self._opt = opt
                self.do_step = False

            def zero_grad(self, set_to_none=True):
                pass

            def step(self):
                if self.do_step:
                    self._opt.step()

        self._ctl = _Ctl(self._opt)

    def progress(self, dataloader_iter):
        if self._i % self._n == 0:
            self._opt.zero_grad(set_to_none=True)  # window start
        self._ctl.do_step = (self._i + 1) % self._n == 0
        self._pipeline._optimizer = self._ctl
        try:
            out = self._pipeline.progress(dataloader_iter)
        finally:
            self._pipeline._optimizer = self._opt
        self._i += 1
        return out


class PipelineStage:
    """User-defined pre-forward stage (reference pipeline_stage.py:74)."""

    def __init__(self, name: str, runnable, stream: Optional[torch.cuda.Stream] = None,
                 fill_callback=None) -> None:
        self.name = name
        self.runnable = runnable
        self.stream = stream
        self.fill_callback = fill_callback


class StagedTrainPipeline:
    """Generic K-stage pipeline: each batch flows through user stages on their
    own streams before the caller's fwd/bwd (reference train_pipelines.py:2579).

    progress() returns the oldest fully-staged batch (or None while filling).
    """

    def __init__(self, pipeline_stages: List[PipelineStage], device: Optional[torch.device] = None) -> None:
        self._stages = pipeline_stages
        self._device = device
        # slot i holds the output of stage i for the batch currently there
        self._slots: List[Optional[Any]] = [None] * (len(pipeline_stages) + 1)
        self._filled = False

    def _run_stage(self, i: int, item: Any) -> Any:
        stage = self._stages[i]
        ctx = (
            torch.cuda.stream(stage.stream)
            if stage.stream is not None
            else contextlib.nullcontext()
        )
        with torch.autograd.profiler.record_function(f"## stage {stage.name} ##"), ctx:
            return stage.runnable(item)

    def progress(self, dataloader_iter: Iterator[Any]) -> Optional[Any]:
        # advance the pipeline one tick: shift each slot forward
        out = self._slots[-1]
        for i in reversed(range(len(self._stages))):
            item = self._slots[i]
            self._slots[i + 1] = self._run_stage(i, item) if item is not None else None
        try:
            self._slots[0] = next(dataloader_iter)
        except StopIteration:
            self._slots[0] = None
        if out is None and any(s is not None for s in self._slots):
            # pipeline still filling
            return self.progress(dataloader_iter) if not self._filled else None
        self._filled = True
        return out


class EvalPipelineSparseDist(TrainPipelineSparseDist):
    """Inference-only variant: no backward/optimizer (reference :2259)."""

    def progress(self, dataloader_iter: Iterator[Pipelineable]) -> Any:
        if not self._batches:
            self._fill_pipeline(dataloader_iter)
        cur_batch = self._batches[0]
        next_batch = self._batches[1]
        if cur_batch is None:
            raise StopIteration
        batch_ip2 = self._copy_batch_to_gpu(dataloader_iter)
        _wait_for_batch(cur_batch, self._memcpy_stream)
        d_next = self._start_sparse_data_dist(next_batch)
        with torch.no_grad(), torch.autograd.profiler.record_function("## forward ##"):
            losses, output = self._model(cur_batch)
        self._batches = [next_batch, batch_ip2]
        self._dist_contexts = [d_next]
        return output


class PrefetchTrainPipelineSparseDist(TrainPipelineSparseDist):
    """4-stage pipeline, 3 batches in flight (reference train_pipelines.py:1967):

    batch i:   fwd/bwd/opt (default stream)
    batch i+1: UVM-cache prefetch (prefetch stream)
    batch i+2: sparse input_dist (data-dist stream)
    batch i+3: H2D copy (memcpy stream)
    """

    def __init__(self, *args, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self._prefetch_stream = (
            torch.cuda.Stream(priority=-1) if self._device.type == "cuda" else None
        )
        self._prefetched_dist: List[Dict[str, Tuple[Any, Any]]] = []

    def _prefetch(self, dists: Dict[str, Tuple[Any, Any]]) -> Dict[str, Tuple[Any, Any]]:
        """Wait the batch's input_dist and run cache prefetch on its stream."""
        out: Dict[str, Tuple[Any, Any]] = {}
        ctxmgr = (
            torch.cuda.stream(self._prefetch_stream)
            if self._prefetch_stream is not None
            else contextlib.nullcontext()
        )
        with torch.autograd.profiler.record_function("## prefetch ##"), ctxmgr:
            for fqn, (ctx, tensors_aw) in dists.items():
                dist_input = tensors_aw.wait()
                module = self._pipelined[fqn]
                for lookup in getattr(module, "_lookups", []):
                    kjts = (
                        dist_input
                        if isinstance(dist_input, list)
                        else [dist_input]
                    )
                    for kjt, tbe_list in zip(kjts, [lookup.tbes()]):
                        for tbe in tbe_list:
                            if getattr(tbe, "_uvm_caching", False):
                                tbe.prefetch(kjt.values(), kjt.offsets())
                out[fqn] = (ctx, _Ready(dist_input))
        return out

    def _fill_pipeline(self, dataloader_iter) -> None:
        b0 = self._copy_batch_to_gpu(dataloader_iter)
        if b0 is None:
            raise StopIteration
        if not self._rewritten:
            self._rewrite_model()
        d0 = self._prefetch(self._start_sparse_data_dist(b0))
        b1 = self._copy_batch_to_gpu(dataloader_iter)
        d1 = self._start_sparse_data_dist(b1)
        b2 = self._copy_batch_to_gpu(dataloader_iter)
        self._batches = [b0, b1, b2]
        self._dist_contexts = [d0, d1]

    def progress(self, dataloader_iter: Iterator[Pipelineable]) -> Any:
        if not self._batches:
            self._fill_pipeline(dataloader_iter)
        cur_batch = self._batches[0]
        if cur_batch is None:
            raise StopIteration
        batch_ip3 = self._copy_batch_to_gpu(dataloader_iter)
        _wait_for_batch(cur_batch, self._memcpy_stream)
        self._optimizer.zero_grad(set_to_none=True)
        # start dist for i+2, prefetch i+1
        d_ip2 = self._start_sparse_data_dist(self._batches[2])
        d_ip1 = self._prefetch(self._dist_contexts[1])
        if self._prefetch_stream is not None:
            torch.cuda.current_stream().wait_stream(self._prefetch_stream)
        ac = (
            torch.autocast(device_type=self._device.type, dtype=self._autocast_dtype)
            if self._autocast_dtype is not None
            else contextlib.nullcontext()
        )
        with torch.autograd.profiler.record_function("## forward ##"), ac:
            losses, output = self._model(cur_batch)
        with torch.autograd.profiler.record_function("## backward ##"):
            torch.sum(losses).backward()
        with torch.autograd.profiler.record_function("## optimizer ##"):
            self._optimizer.step()
        self._batches = [self._batches[1], self._batches[2], batch_ip3]
        self._dist_contexts = [d_ip1, d_ip2]
        return output


class _Ready:
    """Pre-waited awaitable."""

    def __init__(self, value) -> None:
        self._value = value

    def wait(self):
        return self._value


class TrainPipelineSemiSync(TrainPipelineSparseDist):
    """Semi-synchronous pipeline (reference train_pipelines.py:1637): the
    embedding lookup + output_dist for batch i+1 is issued on the lookup
    stream BEFORE batch i's backward/optimizer, so the sparse compute and
    comms hide under the dense backward. Embeddings are therefore one
    optimizer step stale (the documented semi-sync relaxation).
    """

    def __init__(self, *args, stash_gradients: bool = False, **kwargs) -> None:
        super().__init__(*args, **kwargs)
        self._lookup_stream = (
            torch.cuda.Stream(priority=-1) if self._device.type == "cuda" else None
        )
        self._emb_awaitables: List[Dict[str, Any]] = []

    def _start_embedding_lookup(self, dists: Dict[str, Tuple[Any, Any]]) -> Dict[str, Any]:
        out: Dict[str, Any] = {}
        ctxmgr = (
            torch.cuda.stream(self._lookup_stream)
            if self._lookup_stream is not None
            else contextlib.nullcontext()
        )
        with torch.autograd.profiler.record_function("## start_embedding_lookup ##"), ctxmgr:
            for fqn, (ctx, tensors_aw) in dists.items():
                dist_input = tensors_aw.wait()
                module = self._pipelined[fqn]
                out[fqn] = module.compute_and_output_dist(ctx, dist_input)
        return out

    def _fetch_dist(self, fqn: str):  # PipelinedForward hook
        raise RuntimeError("semi-sync consumes precomputed lookups")

    def _rewrite_model(self) -> None:
        for fqn, module in self._model.named_modules():
            if hasattr(module, "compute_and_output_dist") and hasattr(module, "input_dist"):
                self._pipelined[fqn] = module
                pipeline = self

                def make_forward(fq):
                    def fwd(*args, **kwargs):
                        if pipeline._lookup_stream is not None:
                            torch.cuda.current_stream().wait_stream(pipeline._lookup_stream)
                        return pipeline._emb_awaitables[0][fq]

                    return fwd

                module.forward = make_forward(fqn)
        self._rewritten = True

    def _fill_pipeline(self, dataloader_iter) -> None:
        b0 = self._copy_batch_to_gpu(dataloader_iter)
        if b0 is None:
            raise StopIteration
        if not self._rewritten:
            self._rewrite_model()
        e0 = self._start_embedding_lookup(self._start_sparse_data_dist(b0))
        b1 = self._copy_batch_to_gpu(dataloader_iter)
        d1 = self._start_sparse_data_dist(b1)
        self._batches = [b0, b1]
        self._dist_contexts = [d1]
        self._emb_awaitables = [e0]

    def progress(self, dataloader_iter: Iterator[Pipelineable]) -> Any:
        if not self._batches:
            self._fill_pipeline(dataloader_iter)
        cur_batch = self._batches[0]
        next_batch = self._batches[1]
        if cur_batch is None:
            raise StopIteration
        batch_ip2 = self._copy_batch_to_gpu(dataloader_iter)
        _wait_for_batch(cur_batch, self._memcpy_stream)
        self._optimizer.zero_grad(set_to_none=True)
        ac = (
            torch.autocast(device_type=self._device.type, dtype=self._autocast_dtype)
            if self._autocast_dtype is not None
            else contextlib.nullcontext()
        )
        with torch.autograd.profiler.record_function("## forward ##"), ac:
            losses, output = self._model(cur_batch)
        # semi-sync: issue batch i+1's lookup BEFORE batch i's backward
        e_next = (
            self._start_embedding_lookup(self._dist_contexts[0])
            if next_batch is not None
            else {}
        )
        d_ip2 = self._start_sparse_data_dist(batch_ip2)
        with torch.autograd.profiler.record_function("## backward ##"):
            torch.sum(losses).backward()
        with torch.autograd.profiler.record_function("## optimizer ##"):
            self._optimizer.step()
        self._batches = [next_batch, batch_ip2]
        self._dist_contexts = [d_ip2]
        self._emb_awaitables = [e_next]
        return output


class TrainPipelinePT2(TrainPipelineBase):
    """torch.compile pipeline (reference train_pipelines.py:423): the model's
    dense compute is compiled on first progress; sparse structures (KJT)
    enter the graph through the fx-leaf tracer contract, so unsupported
    constructs fall back to graph breaks rather than errors.

    MI355X note: inductor on ROCm emits HIP via triton-cpu/cpp on CPU and is
    primarily useful here for the dense arch; the HIP custom ops stay as
    opaque calls."""

    def __init__(
        self,
        model: torch.nn.Module,
        optimizer: torch.optim.Optimizer,
        device: torch.device,
        autocast_dtype: Optional[torch.dtype] = None,
        compile_kwargs: Optional[dict] = None,
    ) -> None:
        super().__init__(model, optimizer, device, autocast_dtype)
        self._compile_kwargs = dict(compile_kwargs or {})
        self._compiled = False

    def progress(self, dataloader_iter: Iterator[Pipelineable]) -> Any:
        if not self._compiled:
            self._model = torch.compile(self._model, **self._compile_kwargs)
            self._compiled = True
        return super().progress(dataloader_iter)


class MicroBatchPipeline:
    """Micro-batch serving/eval pipeline (reference maglev/pipeline.py:20):
    splits each incoming batch into ``num_micro`` slices and runs them
    back-to-back, overlapping each slice's H2D copy (memcpy stream) with the
    previous slice's compute — bounds serving latency without giving up
    copy/compute overlap."""

    def __init__(self, model, device: torch.device, num_micro: int = 2) -> None:
        self._model = model
        self._device = device
        self._n = max(1, num_micro)
        self._memcpy_stream = (
            torch.cuda.Stream(priority=-1) if device.type == "cuda" else None
        )

    def _split(self, batch):
        if self._n == 1 or not hasattr(batch, "dense_features"):
            return [batch]
        B = batch.dense_features.shape[0]
        step = (B + self._n - 1) // self._n
        out = []
        for lo in range(0, B, step):
            hi = min(B, lo + step)
            kjt = batch.sparse_features
            K = len(kjt.keys())
            lengths2d = kjt.lengths().view(K, B)
            offs = torch.zeros(K * B + 1, dtype=torch.int64)
            torch.cumsum(kjt.lengths(), 0, out=offs[1:])
            vals, lens = [], []
            for k in range(K):
                l0, h0 = int(offs[k * B + lo]), int(offs[k * B + hi])
                vals.append(kjt.values()[l0:h0])
                lens.append(lengths2d[k, lo:hi])
            out.append(
                type(batch)(
                    dense_features=batch.dense_features[lo:hi],
                    sparse_features=KeyedJaggedTensor(
                        keys=kjt.keys(),
                        values=torch.cat(vals),
                        lengths=torch.cat(lens),
                        stride=hi - lo,
                    ),
                    labels=batch.labels[lo:hi],
                )
            )
        return out

    @torch.no_grad()
    def progress(self, batch):
        micro = self._split(batch)
        outs = []
        staged = None
        for i, m in enumerate(micro):
            if self._memcpy_stream is not None:
                with torch.cuda.stream(self._memcpy_stream):
                    nxt = m.to(self._device, non_blocking=True)
            else:
                nxt = m.to(self._device) if self._device.type != "cpu" else m
            if staged is not None:
                outs.append(self._model(staged))
            if self._memcpy_stream is not None:
                torch.cuda.current_stream().wait_stream(self._memcpy_stream)
                nxt.record_stream(torch.cuda.current_stream())
            staged = nxt
        outs.append(self._model(staged))
        first = outs[0]
        if isinstance(first, torch.Tensor):
            return torch.cat(outs)
        return outs
