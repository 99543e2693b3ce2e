"""Mutable shared-memory channels — the compiled-DAG transport.

Reference: src/ray/core_worker/experimental_mutable_object_manager.h:44
(multi-reader/single-writer shm channels with WriteAcquire/ReadAcquire
semaphores) and python/ray/experimental/channel/shared_memory_channel.py.

ray_amd design: one shm file per channel:
  header: [u64 write_seq][u64 msg_size][u64 ack_slot * MAX_READERS]
  body:   payload bytes (capacity fixed at creation)
Single writer bumps write_seq after writing the payload; each reader
owns one ack slot and writes the seq it has consumed. The writer blocks
until all readers have acked the previous message (same backpressure
contract as the reference's WriteAcquire). 8-byte aligned slot writes
are atomic on x86-64.
"""
from __future__ import annotations

import mmap
import os
import struct
import time
from typing import Optional

_HDR = struct.Struct("<QQ")
MAX_READERS = 16
_HDR_SIZE = 16 + 8 * MAX_READERS


class Channel:
    def __init__(self, path: str, capacity: int = 1 << 20,
                 num_readers: int = 1, create: bool = False):
        self.path = path
        self.num_readers = num_readers
        if create:
            fd = os.open(path, os.O_CREAT | os.O_RDWR, 0o600)
            os.ftruncate(fd, _HDR_SIZE + capacity)
        else:
            deadline = time.time() + 30
            while not os.path.exists(path):
                if time.time() > deadline:
                    raise TimeoutError(f"channel {path} never appeared")
                time.sleep(0.005)
            fd = os.open(path, os.O_RDWR)
        size = os.fstat(fd).st_size
        self._mm = mmap.mmap(fd, size)
        os.close(fd)
        self.capacity = size - _HDR_SIZE
        self._view = memoryview(self._mm)

    # ---- header accessors ----

    def _write_seq(self) -> int:
        return _HDR.unpack_from(self._mm, 0)[0]

    def _set(self, seq: int, size: int):
        _HDR.pack_into(self._mm, 0, seq, size)

    def _ack(self, slot: int, seq: int):
        struct.pack_into("<Q", self._mm, 16 + 8 * slot, seq)

    def _min_ack(self) -> int:
        return min(
            struct.unpack_from("<Q", self._mm, 16 + 8 * i)[0]
            for i in range(self.num_readers)
        )

    # ---- writer ----

    def write(self, data: bytes, timeout: Optional[float] = 60.0):
        seq = self._write_seq()
        deadline = None if timeout is None else time.monotonic() + timeout
        while self._min_ack() < seq:  # previous message not fully consumed
            if deadline and time.monotonic() > deadline:
                raise TimeoutError("channel backpressure timeout")
            time.sleep(0.0002)
        n = len(data)
        if n > self.capacity:
            raise ValueError(f"message {n}B exceeds channel capacity")
        self._view[_HDR_SIZE : _HDR_SIZE + n] = data
        self._set(seq + 1, n)

    def write_obj(self, obj, timeout: Optional[float] = 60.0):
        from ray_amd._core import serialization

        self.write(serialization.dumps(obj), timeout)

    # ---- reader ----

    def read(self, slot: int = 0, last_seq: int = 0,
             timeout: Optional[float] = 60.0):
        """Blocks for a seq > last_seq; returns (seq, bytes)."""
        deadline = None if timeout is None else time.monotonic() + timeout
        while True:
            seq, size = _HDR.unpack_from(self._mm, 0)
            if seq > last_seq:
                data = bytes(self._view[_HDR_SIZE : _HDR_SIZE + size])
                self._ack(slot, seq)
                return seq, data
            if deadline and time.monotonic() > deadline:
                raise TimeoutError("channel read timeout")
            time.sleep(0.0002)

    def read_obj(self, slot: int = 0, last_seq: int = 0,
                 timeout: Optional[float] = 60.0):
        from ray_amd._core import serialization

        seq, data = self.read(slot, last_seq, timeout)
        return seq, serialization.loads(data)

    def close(self):
        try:
            self._view.release()
            self._mm.close()
        except Exception:
            pass


class ChannelReader:
    """Stateful reader cursor over a Channel."""

    def __init__(self, channel: Channel, slot: int = 0):
        self.ch = channel
        self.slot = slot
        self.seq = 0

    def next(self, timeout: Optional[float] = 60.0):
        self.seq, data = self.ch.read(self.slot, self.seq, timeout)
        return data

    def next_obj(self, timeout: Optional[float] = 60.0):
        from ray_amd._core import serialization

        return serialization.loads(self.next(timeout))


def channel_path(name: str) -> str:
    """Session-scoped channel file path."""
    from ray_amd._core import runtime as rtmod

    rt = rtmod.global_runtime()
    return os.path.join(rt.shm_dir, f"chan_{name}")
