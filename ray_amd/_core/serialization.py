"""Serialization: cloudpickle + pickle protocol-5 out-of-band buffers.

Counterpart of the reference's SerializationContext
(python/ray/_private/serialization.py:145): values are pickled with a
buffer_callback so large contiguous buffers (numpy arrays, torch CPU
tensors, bytes) are captured out-of-band and written into one shared
memory segment; deserialization reconstructs zero-copy read-only views
onto the mmap. Layout of a sealed object:

  [u32 magic][u32 nbufs][u64 meta_len][u64 buf_len]*nbufs
  [meta (pickle bytes)] [pad to 64] [buf0][pad64][buf1]...
"""
from __future__ import annotations

import pickle
import struct
from typing import Any, List, Tuple

import cloudpickle

MAGIC = 0x52414D44  # "RAMD"
_ALIGN = 64

_PAR_COPY_MIN = 32 * 1024 * 1024
_copy_pool = None


def _parallel_copy(dst: memoryview, src: memoryview):
    """Multi-threaded memcpy — memoryview slice assignment releases the
    GIL, so 4 threads ≈ 2-3x one-thread bandwidth on large buffers."""
    global _copy_pool
    n = src.nbytes
    if n < _PAR_COPY_MIN:
        dst[:] = src
        return
    import concurrent.futures

    if _copy_pool is None:
        _copy_pool = concurrent.futures.ThreadPoolExecutor(
            max_workers=4, thread_name_prefix="shm_copy"
        )
    nthreads = 4
    chunk = (n + nthreads - 1) // nthreads
    futs = []
    for i in range(nthreads):
        s = i * chunk
        e = min(s + chunk, n)
        if s >= e:
            break
        futs.append(_copy_pool.submit(_copy_range, dst, src, s, e))
    for f in futs:
        f.result()


def _copy_range(dst, src, s, e):
    dst[s:e] = src[s:e]

# Threshold below which values are inlined into RPC replies instead of
# the object store (reference: max_direct_call_object_size=100KiB,
# ray_config_def.h:274).
from .._config import config as _cfg

INLINE_MAX = _cfg.inline_max_bytes  # reference: max_direct_call_object_size


def _pad(n: int) -> int:
    return (n + _ALIGN - 1) & ~(_ALIGN - 1)


def serialize(value: Any) -> Tuple[bytes, List[pickle.PickleBuffer]]:
    """Returns (meta, buffers). meta is the pickle stream; buffers are
    out-of-band PickleBuffers (zero-copy views of the original arrays)."""
    buffers: List[pickle.PickleBuffer] = []
    meta = cloudpickle.dumps(value, protocol=5, buffer_callback=buffers.append)
    return meta, buffers


def serialized_size(meta: bytes, buffers) -> int:
    total = _pad(16 + 8 * len(buffers) + len(meta))
    for b in buffers:
        total += _pad(b.raw().nbytes)
    return total


def write_to(buf: memoryview, meta: bytes, buffers) -> int:
    """Write the object into a writable memoryview; returns bytes used."""
    nbufs = len(buffers)
    struct.pack_into("<IIQ", buf, 0, MAGIC, nbufs, len(meta))
    off = 16
    for b in buffers:
        struct.pack_into("<Q", buf, off, b.raw().nbytes)
        off += 8
    mlen = len(meta)
    buf[off : off + mlen] = meta
    off = _pad(off + mlen)
    for b in buffers:
        raw = b.raw()
        n = raw.nbytes
        src = raw.cast("B") if raw.format != "B" or raw.ndim != 1 else raw
        _parallel_copy(buf[off : off + n], src)
        off = _pad(off + n)
    return off


def dumps(value: Any) -> bytes:
    """One-shot in-band serialization (for RPC-inlined values)."""
    meta, buffers = serialize(value)
    out = bytearray(serialized_size(meta, buffers))
    n = write_to(memoryview(out), meta, buffers)
    return bytes(out[:n])


def loads_from(buf: memoryview) -> Any:
    """Zero-copy deserialize from a (possibly mmap'd) buffer."""
    magic, nbufs, mlen = struct.unpack_from("<IIQ", buf, 0)
    if magic != MAGIC:
        raise ValueError("corrupt object header")
    off = 16
    sizes = []
    for _ in range(nbufs):
        (s,) = struct.unpack_from("<Q", buf, off)
        sizes.append(s)
        off += 8
    meta = bytes(buf[off : off + mlen])
    off = _pad(off + mlen)
    views = []
    for s in sizes:
        views.append(buf[off : off + s])
        off = _pad(off + s)
    return pickle.loads(meta, buffers=views)


def loads(data: bytes) -> Any:
    return loads_from(memoryview(data))
