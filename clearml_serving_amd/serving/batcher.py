"""Dynamic request batcher: coalesce HTTP requests into GPU batches.

Replaces Triton's server-side ``dynamic_batching`` (reference pass-through
config: preferred_batch_size [1,2,4,8,16,32,64], max_batch_size 64,
max_queue_delay, examples/huggingface/readme.md:113) with a native asyncio
coalescer in the serving process:

- per-endpoint queue; the first request opens a batching window of
  ``max_queue_delay_us``; the window closes early when ``max_batch_size``
  requests arrived.
- batches are padded up to the next *bucket* size so steady-state shapes are
  stable, which lets each bucket be captured once into a hipGraph
  (torch.cuda.CUDAGraph == hipGraph on ROCm) and replayed afterwards --
  launch overhead drops from one-kernel-at-a-time to a single graph launch.
- model execution runs on a dedicated HIP stream per batcher in a worker
  thread, so the asyncio loop keeps accepting requests while the GPU runs;
  D2H copies of the outputs synchronize only that stream.

MI355X sizing note: with 288 GB HBM3E per GPU the default buckets are capped
by latency targets, not memory -- raise max_batch_size freely via
``auxiliary_cfg`` for throughput endpoints.
"""

import asyncio
import queue as _queue
import threading
import time
from typing import Callable, Dict, List, Optional, Sequence, Union

import torch

DEFAULT_BUCKETS = (1, 2, 4, 8, 16, 32, 64)

# hipGraph capture is process-global state: serialize captures across batchers
_capture_lock = threading.Lock()

TensorOrDict = Union[torch.Tensor, Dict[str, torch.Tensor]]

# staging memcpy pool: torch's CPU copy for bf16 runs element-wise on the
# serving box (~1 GB/s measured -- 17 ms to stack a 19 MB ResNet batch);
# numpy byte-view assignment is a true memcpy (6 GB/s single thread, 34 GB/s
# across 8) -- scripts/pinprobe.py has the measurement
_stage_pool = None
_stage_pool_lock = threading.Lock()
_PAR_STAGE_BYTES = 2 << 20  # parallelize copies above 2 MiB total


def _stage_executor():
    global _stage_pool
    if _stage_pool is None:
        with _stage_pool_lock:
            if _stage_pool is None:
                from concurrent.futures import ThreadPoolExecutor

                _stage_pool = ThreadPoolExecutor(
                    max_workers=8, thread_name_prefix="cmls-stage")
    return _stage_pool


def _numpy_stack_into(buf: torch.Tensor, ts: List[torch.Tensor],
                      bucket: int) -> None:
    """Copy ``ts`` into rows [0, n) of the (pinned) slab and replicate row 0
    into the pad rows, via numpy byte views (real memcpy, GIL released)."""
    n = len(ts)
    dst = buf[:bucket].view(torch.uint8).numpy()
    srcs = [t.contiguous().view(torch.uint8).numpy().reshape(-1) for t in ts]
    total = n * srcs[0].size

    def copy_range(lo, hi):
        for i in range(lo, hi):
            dst[i] = srcs[i].reshape(dst[i].shape)

    if total >= _PAR_STAGE_BYTES and n >= 8:
        pool = _stage_executor()
        nth = min(8, n)
        chunk = (n + nth - 1) // nth
        futs = [pool.submit(copy_range, lo, min(lo + chunk, n))
                for lo in range(0, n, chunk)]
        for f in futs:
            f.result()
    else:
        copy_range(0, n)
    if n < bucket:
        dst[n:bucket] = dst[0]


class DynamicBatcher:
    def __init__(
        self,
        model_fn: Callable[[TensorOrDict], TensorOrDict],
        device: Union[str, torch.device] = "cuda",
        max_batch_size: int = 64,
        max_queue_delay_us: int = 2000,
        preferred_batch_sizes: Sequence[int] = DEFAULT_BUCKETS,
        use_graphs: bool = True,
        dtype: Optional[torch.dtype] = None,
        name: str = "endpoint",
        out_convert: Optional[Callable] = None,
    ):
        self._model_fn = model_fn
        self.device = torch.device(device)
        self.is_cuda = self.device.type == "cuda"
        self.max_batch_size = int(max_batch_size)
        self.max_queue_delay_s = float(max_queue_delay_us) / 1e6
        self.buckets = sorted(
            {min(b, self.max_batch_size) for b in preferred_batch_sizes}
            | {self.max_batch_size}
        )
        self.use_graphs = bool(use_graphs) and self.is_cuda
        self.dtype = dtype
        self.name = name
        # applied ONCE to the whole batched output in the worker thread
        # (e.g. bf16 -> float numpy); per-request conversion was 512 small
        # torch ops per bench step on the event loop. When set, the split
        # returns views into the converted (freshly materialized) batch.
        self.out_convert = out_convert

        self._queue: "asyncio.Queue" = None  # created lazily on the loop
        self._loop: Optional[asyncio.AbstractEventLoop] = None
        self._worker_task: Optional[asyncio.Task] = None
        self._stream = torch.cuda.Stream() if self.is_cuda else None
        self._graphs: Dict[int, dict] = {}  # bucket -> {graph, in, out}
        self._pinned: Dict = {}             # (slot,key,bucket,...) -> slab
        self._exec_lock = threading.Lock()  # CPU path only
        # pipelining: up to 2 batches in flight (CPU staging of batch n+1
        # overlaps GPU execution of batch n); the enqueue lock makes each
        # batch's GPU ops contiguous on the stream, so the in-order stream
        # keeps static graph buffers race-free
        self._slots: "_queue.Queue" = _queue.Queue()
        import os as _os

        for s in range(int(_os.environ.get("CMLS_BATCH_SLOTS", 3))):
            self._slots.put(s)
        self._enqueue_lock = threading.Lock()
        self._inflight = set()
        self._closed = False
        # telemetry for the Prometheus exporter / GET /status
        self.stats = {"batches": 0, "requests": 0, "occupancy_sum": 0.0,
                      "stage_ms_sum": 0.0, "gpu_wait_ms_sum": 0.0,
                      "queue_wait_ms_sum": 0.0}

    # ------------------------------------------------------------------ #
    async def submit(self, inputs: TensorOrDict) -> TensorOrDict:
        """Enqueue one request's input tensors (no batch dim); returns this
        request's slice of the model output."""
        self._ensure_worker()
        fut = asyncio.get_running_loop().create_future()
        await self._queue.put((inputs, fut, time.monotonic()))
        return await fut

    def _ensure_worker(self) -> None:
        loop = asyncio.get_running_loop()
        if (self._worker_task is None or self._loop is not loop
                or self._worker_task.done()):
            # .done() guard: restart a worker that died on an unexpected
            # exception (mirrors the LLM engine-loop restart) -- otherwise
            # every future queued after the crash would hang forever
            self._loop = loop
            self._queue = asyncio.Queue()
            self._worker_task = loop.create_task(self._worker())

    async def close(self) -> None:
        self._closed = True
        if self._worker_task:
            self._worker_task.cancel()
            self._worker_task = None

    def shutdown(self) -> None:
        """Thread-safe teardown (called when the endpoint is removed on a
        config reload): stop the worker and drop graphs/slabs so HBM frees
        with the engine instance."""
        self._closed = True
        task, loop = self._worker_task, self._loop
        self._worker_task = None
        if task is not None and loop is not None and not loop.is_closed():
            loop.call_soon_threadsafe(task.cancel)
        self._graphs.clear()
        self._pinned.clear()

    # ------------------------------------------------------------------ #
    async def _worker(self) -> None:
        while not self._closed:
            first = await self._queue.get()
            batch: List = [first]
            deadline = time.monotonic() + self.max_queue_delay_s
            while len(batch) < self.max_batch_size:
                # drain whatever is already queued without timer machinery
                try:
                    while len(batch) < self.max_batch_size:
                        batch.append(self._queue.get_nowait())
                    break
                except asyncio.QueueEmpty:
                    pass
                timeout = deadline - time.monotonic()
                if timeout <= 0:
                    break
                try:
                    item = await asyncio.wait_for(self._queue.get(), timeout)
                    batch.append(item)
                except asyncio.TimeoutError:
                    break
            now = time.monotonic()
            self.stats["queue_wait_ms_sum"] += sum(
                (now - b[2]) * 1000 for b in batch)
            # group by input signature (keys/shapes/dtypes): a window can
            # coalesce heterogeneous requests, and stacking them together
            # would crash the WHOLE batch -- innocent co-batched requests
            # included. Homogeneous traffic stays one group.
            groups: Dict = {}
            for inp, fut, t0 in batch:
                groups.setdefault(self._signature(inp), []).append(
                    (inp, fut))
            for items in groups.values():
                inputs = [it[0] for it in items]
                futures = [it[1] for it in items]
                # fire-and-continue: the next batch forms while this one
                # stages and executes (in-flight bounded by the slot
                # queue); hold a reference so the task isn't GC'd mid-run
                task = asyncio.get_running_loop().create_task(
                    self._dispatch(inputs, futures))
                self._inflight.add(task)
                task.add_done_callback(self._inflight.discard)

    @staticmethod
    def _signature(x):
        # hot loop (once per request): torch.dtype/shape tuples are
        # hashable as-is -- no string formatting
        try:
            if isinstance(x, dict):
                return tuple(sorted(
                    (k, tuple(v.shape), v.dtype)
                    for k, v in x.items()))
            return (tuple(x.shape), x.dtype)
        except Exception:
            return ("opaque", id(type(x)))

    async def _dispatch(self, inputs, futures) -> None:
        try:
            outputs = await asyncio.to_thread(self._execute, inputs)
            for fut, out in zip(futures, outputs):
                if not fut.done():
                    fut.set_result(out)
        except Exception as ex:
            for fut in futures:
                if not fut.done():
                    fut.set_exception(ex)

    def warmup(self, sample: TensorOrDict) -> None:
        """Pre-capture every bucket's hipGraph from one sample request so no
        capture happens on the serving path (call at endpoint creation)."""
        if not self.use_graphs:
            return
        for bucket in self.buckets:
            self._execute([sample] * bucket)

    # ------------------------------------------------------------------ #
    def _execute(self, inputs: List[TensorOrDict]) -> List[TensorOrDict]:
        """Assemble the batch, run on the batcher's HIP stream, slice back."""
        n = len(inputs)
        bucket = next(b for b in self.buckets if b >= n)
        self.stats["batches"] += 1
        self.stats["requests"] += n
        self.stats["occupancy_sum"] += n / float(bucket)

        is_dict = isinstance(inputs[0], dict)

        if not self.is_cuda:
            def _stage_cpu():
                if is_dict:
                    keys = list(inputs[0].keys())
                    return {k: self._stack_pad([x[k] for x in inputs],
                                               bucket, key=k) for k in keys}
                return self._stack_pad(inputs, bucket)

            with self._exec_lock:
                out_cpu = _to_cpu(self._run_model(_stage_cpu(), bucket))
            if self.out_convert is not None:
                out_cpu = self.out_convert(out_cpu)
            return [_slice(out_cpu, i) for i in range(n)]

        slot = self._slots.get()  # bounds in-flight batches (2)
        try:
            t_stage = time.monotonic()
            # CPU staging into this slot's pinned slab overlaps the other
            # slot's GPU execution
            if is_dict:
                staged = {k: self._stack_pad([x[k] for x in inputs], bucket,
                                             key=k, slot=slot)
                          for k in inputs[0].keys()}
            else:
                staged = self._stack_pad(inputs, bucket, slot=slot)

            with self._enqueue_lock:
                with torch.cuda.stream(self._stream):
                    dev = _to_device(staged, self.device)
                    out = self._run_model(dev, bucket)
                    out_cpu = self._pinned_out(out, slot)
                done = torch.cuda.Event()
                done.record(self._stream)
            t_enq = time.monotonic()
            done.synchronize()
            t_done = time.monotonic()
            self.stats["stage_ms_sum"] += (t_enq - t_stage) * 1000
            self.stats["gpu_wait_ms_sum"] += (t_done - t_enq) * 1000
            if self.out_convert is not None:
                # conversion materializes a fresh batch: views are safe
                conv = self.out_convert(out_cpu)
                return [_slice(conv, i) for i in range(n)]
            # slices are views into the reusable pinned slab: clone them
            return [_slice_clone(out_cpu, i) for i in range(n)]
        finally:
            self._slots.put(slot)

    def _stack_pad(self, tensors: List[torch.Tensor], bucket: int,
                   key: str = "", slot: int = 0) -> torch.Tensor:
        ts = [torch.as_tensor(x) for x in tensors]
        want = self.dtype if (self.dtype is not None
                              and ts[0].is_floating_point()) else ts[0].dtype
        if ts[0].dtype != want:
            ts = [t.to(want) for t in ts]
        if self.is_cuda:
            if ts[0].is_cuda:
                # GPU-resident inputs (user preprocess ran on device): stack
                # on device, skip the pinned host staging entirely
                t = torch.stack(ts, dim=0)
                if t.shape[0] < bucket:
                    pad = t[:1].expand(bucket - t.shape[0], *t.shape[1:])
                    t = torch.cat([t, pad], dim=0)
                return t
            # fill this slot's reusable pinned host slab (the H2D copy is
            # enqueued later on the stream, inside the enqueue lock); one
            # max-batch slab per (slot, input) serves every bucket
            buf = self._pinned_slab(("in", slot, key), self.buckets[-1],
                                    ts[0].shape, want)
            n = len(ts)
            _numpy_stack_into(buf, ts, bucket)
            return buf[:bucket]
        t = torch.stack(ts, dim=0)
        if t.shape[0] < bucket:
            pad = t[:1].expand(bucket - t.shape[0], *t.shape[1:])
            t = torch.cat([t, pad], dim=0)
        return t

    def _pinned_slab(self, key, bucket: int, shape, dtype) -> torch.Tensor:
        slab_key = (key, bucket, tuple(shape), dtype)
        slab = self._pinned.get(slab_key)
        if slab is None:
            slab = torch.empty((bucket, *shape), dtype=dtype,
                               pin_memory=True)
            self._pinned[slab_key] = slab
        return slab

    def _pinned_out(self, out, slot):
        # async D2H into this slot's pinned output slab(s)
        if isinstance(out, dict):
            return {k: self._pinned_out(v, (slot, k)) for k, v in out.items()}
        if isinstance(out, (tuple, list)):
            return type(out)(self._pinned_out(v, (slot, i))
                             for i, v in enumerate(out))
        buf = self._pinned_slab(("out", slot), out.shape[0], out.shape[1:],
                                out.dtype)
        buf.copy_(out, non_blocking=True)
        return buf

    @torch.inference_mode()
    def _run_model(self, stacked: TensorOrDict, bucket: int) -> TensorOrDict:
        if not self.use_graphs:
            return self._model_fn(stacked)

        entry = self._graphs.get(bucket)
        if entry is None:
            with _capture_lock:
                entry = self._capture(stacked, bucket)
            self._graphs[bucket] = entry
        # copy inputs into the graph's static buffers, replay, read outputs
        _copy_into(entry["in"], stacked)
        entry["graph"].replay()
        return entry["out"]

    def _capture(self, stacked: TensorOrDict, bucket: int) -> dict:
        static_in = _clone(stacked)
        # warmup on a side stream (required before capture)
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(3):
                out = self._model_fn(static_in)
        torch.cuda.current_stream().wait_stream(s)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph, stream=torch.cuda.current_stream()):
            static_out = self._model_fn(static_in)
        return {"graph": graph, "in": static_in, "out": static_out}


def _clone(x: TensorOrDict) -> TensorOrDict:
    if isinstance(x, dict):
        return {k: v.clone() for k, v in x.items()}
    return x.clone()


def _copy_into(dst: TensorOrDict, src: TensorOrDict) -> None:
    if isinstance(dst, dict):
        for k in dst:
            dst[k].copy_(src[k], non_blocking=True)
    else:
        dst.copy_(src, non_blocking=True)


def _to_cpu(x) -> TensorOrDict:
    if isinstance(x, dict):
        return {k: v.detach().to("cpu", non_blocking=True) for k, v in x.items()}
    if isinstance(x, (tuple, list)):
        return type(x)(_to_cpu(v) for v in x)
    return x.detach().to("cpu", non_blocking=True)


def _slice(x, i: int):
    if isinstance(x, dict):
        return {k: v[i] for k, v in x.items()}
    if isinstance(x, (tuple, list)):
        return type(x)(_slice(v, i) for v in x)
    return x[i]


def _slice_clone(x, i: int):
    if isinstance(x, dict):
        return {k: _slice_clone(v, i) for k, v in x.items()}
    if isinstance(x, (tuple, list)):
        return type(x)(_slice_clone(v, i) for v in x)
    return x[i].clone()


def _to_device(x, device):
    if isinstance(x, dict):
        return {k: _to_device(v, device) for k, v in x.items()}
    if isinstance(x, (tuple, list)):
        return type(x)(_to_device(v, device) for v in x)
    return x.to(device, non_blocking=True)
