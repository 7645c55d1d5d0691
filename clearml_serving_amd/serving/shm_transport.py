"""Shared-memory transport between HTTP front workers and the engine-owner
process that exclusively owns each GPU.

Why: N uvicorn workers with N engine copies split the dynamic batches and
LOSE throughput on GPU endpoints (measured: profiles/README.md §5), while a
single Python HTTP process caps at ~600 req/s of connection+JSON handling.
The fix is the topology the reference cannot express (it delegates GPU work
to a Triton *container* over gRPC, preprocess_service.py:313-446): many HTTP
front processes parse/preprocess, then hand tensors to ONE engine owner per
GPU over SPSC shared-memory rings, so batches stay whole.

Message framing (little-endian):
  request:  u64 req_id | u16 url_len | url | u8 kind | body
  response: u64 req_id | u8 status   | body
    kind/status 0: tensor payload  -- u16 n; per tensor:
        u16 name_len | name | u8 dtype_code | u8 ndim | u32*ndim | u64 nbytes | raw
    kind/status 1: pickled python object (fallback for non-tensor data)
    status 2: error -- utf-8 message
"""

import asyncio
import itertools
import pickle
import struct
from typing import Any, Dict, List, Optional, Tuple

import numpy as np

try:
    from ._shmring import ShmRing  # C++ ring (built by __graft_entry__)
    HAVE_NATIVE_RING = True
except ImportError:
    ShmRing = None
    HAVE_NATIVE_RING = False


# --------------------------------------------------------------------- #
# pure-Python fallback ring (same record format/semantics as the C++ one;
# used when the native extension has not been built yet)
# --------------------------------------------------------------------- #
class PyShmRing:
    _HDR = 192
    _WRAP = 0xFFFFFFFF

    def __init__(self, name: str, capacity: int, create: bool):
        from multiprocessing import shared_memory

        capacity = (capacity + 7) & ~7
        if create:
            try:
                self._shm = shared_memory.SharedMemory(
                    name=name.lstrip("/"), create=True,
                    size=self._HDR + capacity)
            except FileExistsError:
                shared_memory.SharedMemory(name=name.lstrip("/")).unlink()
                self._shm = shared_memory.SharedMemory(
                    name=name.lstrip("/"), create=True,
                    size=self._HDR + capacity)
            self._buf = self._shm.buf
            struct.pack_into("<QQ", self._buf, 0, 0xC0FFEE, capacity)
            struct.pack_into("<Q", self._buf, 64, 0)   # head
            struct.pack_into("<Q", self._buf, 128, 0)  # tail
        else:
            self._shm = shared_memory.SharedMemory(name=name.lstrip("/"))
            self._buf = self._shm.buf
        self._cap = struct.unpack_from("<Q", self._buf, 8)[0]
        self._data_off = self._HDR
        # opt out of resource_tracker unlink-on-exit double-free noise
        try:
            from multiprocessing import resource_tracker

            resource_tracker.unregister(self._shm._name, "shared_memory")
        except Exception:
            pass

    def _head(self):
        return struct.unpack_from("<Q", self._buf, 64)[0]

    def _tail(self):
        return struct.unpack_from("<Q", self._buf, 128)[0]

    def push(self, data: bytes) -> bool:
        ln = len(data)
        need = (4 + ln + 7) & ~7
        # same guard as the C++ ring: a record over cap/2 can deadlock at
        # an unlucky wrap position (at_end + need > cap forever)
        if need > self._cap // 2:
            raise RuntimeError("record larger than half the ring capacity "
                               "(raise --ring-mb)")
        head, tail = self._head(), self._tail()
        pos = head % self._cap
        at_end = self._cap - pos
        total = need if at_end >= need else at_end + need
        if self._cap - (head - tail) < total:
            return False
        if at_end < need:
            if at_end >= 4:
                struct.pack_into("<I", self._buf, self._data_off + pos,
                                 self._WRAP)
            head += at_end
            pos = 0
        struct.pack_into("<I", self._buf, self._data_off + pos, ln)
        self._buf[self._data_off + pos + 4:
                  self._data_off + pos + 4 + ln] = data
        struct.pack_into("<Q", self._buf, 64, head + need)
        return True

    def drain(self, max_n: int = 1024) -> List[bytes]:
        out = []
        head, tail = self._head(), self._tail()
        while tail < head and len(out) < max_n:
            pos = tail % self._cap
            at_end = self._cap - pos
            if at_end < 4:
                tail += at_end
                continue
            ln = struct.unpack_from("<I", self._buf, self._data_off + pos)[0]
            if ln == self._WRAP:
                tail += at_end
                continue
            out.append(bytes(self._buf[self._data_off + pos + 4:
                                       self._data_off + pos + 4 + ln]))
            tail += (4 + ln + 7) & ~7
        struct.pack_into("<Q", self._buf, 128, tail)
        return out

    def pending(self) -> int:
        return self._head() - self._tail()

    def capacity(self) -> int:
        return self._cap

    def close(self):
        try:
            self._buf = None
            self._shm.close()
        except Exception:
            pass

    @staticmethod
    def unlink(name: str):
        from multiprocessing import shared_memory

        try:
            shm = shared_memory.SharedMemory(name=name.lstrip("/"))
            shm.close()
            shm.unlink()
        except FileNotFoundError:
            pass


def make_ring(name: str, capacity: int, create: bool):
    if HAVE_NATIVE_RING:
        return ShmRing(name, capacity, create)
    return PyShmRing(name, capacity, create)


def unlink_ring(name: str):
    if HAVE_NATIVE_RING:
        ShmRing.unlink(name)
    else:
        PyShmRing.unlink(name)


# --------------------------------------------------------------------- #
# tensor payload framing
# --------------------------------------------------------------------- #
_DTYPE_CODES = {
    np.dtype("float32"): 0, np.dtype("float64"): 1, np.dtype("int64"): 2,
    np.dtype("int32"): 3, np.dtype("int16"): 4, np.dtype("int8"): 5,
    np.dtype("uint8"): 6, np.dtype("bool"): 7, np.dtype("float16"): 8,
}
_CODE_DTYPES = {v: k for k, v in _DTYPE_CODES.items()}

KIND_TENSORS = 0
KIND_PICKLE = 1
KIND_ABORT = 2   # client gone: cancel the in-flight request (req_id only)
STATUS_OK_TENSORS = 0
STATUS_OK_PICKLE = 1
STATUS_ERROR = 2
STATUS_STREAM_CHUNK = 3  # SSE chunk bytes; more follow
STATUS_STREAM_END = 4    # stream finished


def _as_numpy(v) -> Optional[np.ndarray]:
    if isinstance(v, np.ndarray):
        return v if v.dtype in _DTYPE_CODES else None
    try:
        import torch

        if isinstance(v, torch.Tensor):
            if v.dtype in (torch.bfloat16, torch.float16):
                v = v.float()
            a = v.detach().cpu().numpy()
            return a if a.dtype in _DTYPE_CODES else None
    except ImportError:
        pass
    if isinstance(v, (list, tuple)):
        try:
            a = np.asarray(v)
            if a.dtype == np.float64:
                with np.errstate(over="ignore"):
                    a32 = a.astype(np.float32)
                # keep float64 when the downcast would overflow finite
                # values to inf (serving inputs are float32-scale, but a
                # silent inf would corrupt the request)
                if np.all(np.isfinite(a32) | ~np.isfinite(a)):
                    a = a32
            return a if a.dtype in _DTYPE_CODES else None
        except Exception:
            return None
    return None


def _tensor_map(data) -> Optional[List[Tuple[str, np.ndarray]]]:
    """Normalize request data to [(name, array)]; '' name = positional."""
    if isinstance(data, dict):
        pairs = []
        for k, v in data.items():
            a = _as_numpy(v)
            if a is None:
                return None
            pairs.append((str(k), a))
        return pairs
    a = _as_numpy(data)
    if a is None:
        return None
    return [("", a)]


def pack_tensors(pairs: List[Tuple[str, np.ndarray]]) -> bytes:
    parts = [struct.pack("<H", len(pairs))]
    for name, a in pairs:
        nb = name.encode()
        a = np.ascontiguousarray(a)
        raw = a.tobytes()
        parts.append(struct.pack("<H", len(nb)))
        parts.append(nb)
        parts.append(struct.pack("<BB", _DTYPE_CODES[a.dtype], a.ndim))
        parts.append(struct.pack("<{}I".format(a.ndim), *a.shape))
        parts.append(struct.pack("<Q", len(raw)))
        parts.append(raw)
    return b"".join(parts)


def unpack_tensors(buf: bytes, off: int = 0):
    (n,) = struct.unpack_from("<H", buf, off)
    off += 2
    pairs = []
    for _ in range(n):
        (nl,) = struct.unpack_from("<H", buf, off)
        off += 2
        name = buf[off:off + nl].decode()
        off += nl
        code, ndim = struct.unpack_from("<BB", buf, off)
        off += 2
        shape = struct.unpack_from("<{}I".format(ndim), buf, off)
        off += 4 * ndim
        (nbytes,) = struct.unpack_from("<Q", buf, off)
        off += 8
        a = np.frombuffer(buf[off:off + nbytes],
                          dtype=_CODE_DTYPES[code]).reshape(shape)
        off += nbytes
        pairs.append((name, a))
    return pairs


def pack_request(req_id: int, url: str, data: Any) -> bytes:
    ub = url.encode()
    head = struct.pack("<QH", req_id, len(ub)) + ub
    pairs = _tensor_map(data)
    if pairs is not None:
        return head + bytes([KIND_TENSORS]) + pack_tensors(pairs)
    return head + bytes([KIND_PICKLE]) + pickle.dumps(data, protocol=4)


def unpack_request(buf: bytes) -> Tuple[int, str, Any]:
    req_id, ulen = struct.unpack_from("<QH", buf, 0)
    off = 10
    url = buf[off:off + ulen].decode()
    off += ulen
    kind = buf[off]
    off += 1
    if kind == KIND_ABORT:
        return req_id, None, None
    if kind == KIND_TENSORS:
        pairs = unpack_tensors(buf, off)
        if len(pairs) == 1 and pairs[0][0] == "":
            return req_id, url, pairs[0][1]
        return req_id, url, {k: v for k, v in pairs}
    return req_id, url, pickle.loads(buf[off:])


def pack_response(req_id: int, result: Any = None,
                  error: Optional[str] = None) -> bytes:
    head = struct.pack("<Q", req_id)
    if error is not None:
        return head + bytes([STATUS_ERROR]) + error.encode()
    pairs = _tensor_map(result)
    if pairs is not None:
        return head + bytes([STATUS_OK_TENSORS]) + pack_tensors(pairs)
    return head + bytes([STATUS_OK_PICKLE]) + pickle.dumps(result, protocol=4)


def pack_abort(req_id: int) -> bytes:
    return struct.pack("<QH", req_id, 0) + bytes([KIND_ABORT])


def pack_stream_chunk(req_id: int, chunk: bytes) -> bytes:
    return struct.pack("<Q", req_id) + bytes([STATUS_STREAM_CHUNK]) + chunk


def pack_stream_end(req_id: int) -> bytes:
    return struct.pack("<Q", req_id) + bytes([STATUS_STREAM_END])


def unpack_response(buf: bytes) -> Tuple[int, int, Any]:
    (req_id,) = struct.unpack_from("<Q", buf, 0)
    status = buf[8]
    off = 9
    if status in (STATUS_STREAM_CHUNK, STATUS_STREAM_END):
        return req_id, status, bytes(buf[off:])
    if status == STATUS_OK_TENSORS:
        pairs = unpack_tensors(buf, off)
        if len(pairs) == 1 and pairs[0][0] == "":
            return req_id, status, pairs[0][1]
        return req_id, status, {k: v for k, v in pairs}
    if status == STATUS_OK_PICKLE:
        return req_id, status, pickle.loads(buf[off:])
    return req_id, status, buf[off:].decode()


# --------------------------------------------------------------------- #
# front-side client: one per front worker process
# --------------------------------------------------------------------- #
class ShmClient:
    """Front worker's connection to the engine owners.

    ``infer(url, data)`` packs, pushes to the owner's request ring and
    awaits the response future; a single poller task drains the response
    ring and resolves futures by req_id."""

    def __init__(self, prefix: str, worker_id: int, n_owners: int = 1,
                 ring_bytes: int = 32 << 20):
        self.worker_id = worker_id
        self.req_rings = [
            make_ring("{}_req_{}_{}".format(prefix, o, worker_id),
                      ring_bytes, False)
            for o in range(n_owners)
        ]
        self.resp_rings = [
            make_ring("{}_resp_{}_{}".format(prefix, o, worker_id),
                      ring_bytes, False)
            for o in range(n_owners)
        ]
        self._futures: Dict[int, asyncio.Future] = {}
        self._streams: Dict[int, "asyncio.Queue"] = {}
        self._next_id = itertools.count(1)
        self._poller: Optional[asyncio.Task] = None

    def _ensure_poller(self):
        loop = asyncio.get_running_loop()
        if self._poller is None or self._poller.done() \
                or getattr(self, "_loop", None) is not loop:
            self._loop = loop
            self._poller = loop.create_task(self._poll())

    async def _poll(self):
        idle_sleep = 0.0002
        while True:
            got = False
            for ring in self.resp_rings:
                for raw in ring.drain(512):
                    got = True
                    try:
                        req_id, status, payload = unpack_response(raw)
                    except Exception:  # malformed record must not kill the
                        import traceback  # poller (every future would hang)

                        traceback.print_exc()
                        continue
                    if status in (STATUS_STREAM_CHUNK, STATUS_STREAM_END):
                        q = self._streams.get(req_id)
                        if q is not None:
                            q.put_nowait((status, payload))
                        continue
                    if status == STATUS_ERROR and req_id in self._streams:
                        # mid-stream failure routes to the stream consumer
                        self._streams[req_id].put_nowait((status, payload))
                        continue
                    fut = self._futures.pop(req_id, None)
                    if fut is not None and not fut.done():
                        fut.set_result((status, payload))
            if not got:
                await asyncio.sleep(idle_sleep)
            else:
                await asyncio.sleep(0)

    async def infer(self, url: str, data: Any, owner: int = 0,
                    timeout: float = 120.0) -> Any:
        self._ensure_poller()
        req_id = next(self._next_id)
        fut = asyncio.get_running_loop().create_future()
        self._futures[req_id] = fut
        payload = pack_request(req_id, url, data)
        ring = self.req_rings[owner]
        try:
            while not ring.push(payload):
                await asyncio.sleep(0.001)  # ring full: backpressure
        except asyncio.CancelledError:
            # cancelled during backpressure: nothing reached the owner
            self._futures.pop(req_id, None)
            raise
        try:
            status, result = await asyncio.wait_for(fut, timeout=timeout)
        except asyncio.TimeoutError:
            self._futures.pop(req_id, None)
            raise RuntimeError(
                "engine owner did not answer within {}s".format(timeout))
        except asyncio.CancelledError:
            # client disconnected mid-request: tell the owner to stop
            # computing (LLM generations would otherwise run to
            # max_tokens for a dead client)
            self._futures.pop(req_id, None)
            try:
                ring.push(pack_abort(req_id))
            except Exception:
                pass
            raise
        if status == STATUS_ERROR:
            raise RuntimeError(result)
        return result

    async def infer_stream(self, url: str, data: Any, owner: int = 0,
                           timeout: float = 300.0):
        """Streaming variant: yields raw SSE chunk bytes as the engine
        owner relays them (STATUS_STREAM_CHUNK records on the response
        ring, which is FIFO -- chunk order is preserved)."""
        self._ensure_poller()
        req_id = next(self._next_id)
        q: "asyncio.Queue" = asyncio.Queue()
        self._streams[req_id] = q
        try:
            payload = pack_request(req_id, url, data)
            ring = self.req_rings[owner]
            while not ring.push(payload):
                await asyncio.sleep(0.001)
            while True:
                status, chunk = await asyncio.wait_for(q.get(),
                                                       timeout=timeout)
                if status == STATUS_STREAM_END:
                    return
                if status == STATUS_ERROR:
                    raise RuntimeError(chunk if isinstance(chunk, str)
                                       else chunk.decode())
                yield chunk
        finally:
            self._streams.pop(req_id, None)
            # client gone mid-stream: tell the owner to cancel generation
            # (best effort; the owner otherwise runs to max_tokens)
            try:
                ring.push(pack_abort(req_id))
            except Exception:
                pass

    def close(self):
        if self._poller is not None:
            self._poller.cancel()
        for r in self.req_rings + self.resp_rings:
            r.close()
