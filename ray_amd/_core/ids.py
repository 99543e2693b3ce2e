"""Binary IDs for ray_amd.

Design (MI355X-native build; feature parity with reference
src/ray/common/id.h + design_docs/id_specification.md, re-designed):
all ids are fixed-size random byte strings with a readable hex() form.
ObjectIds embed the owner's 8-byte worker nonce + a 64-bit sequence so
an owner can mint ids without coordination (same property the reference
gets by embedding TaskID+index, id.h).
"""
from __future__ import annotations

import os
import struct
import threading

OBJECT_ID_LEN = 16
ACTOR_ID_LEN = 12
TASK_ID_LEN = 12
NODE_ID_LEN = 12
PG_ID_LEN = 12


def _rand(n: int) -> bytes:
    return os.urandom(n)


class _Seq:
    __slots__ = ("v", "lock")

    def __init__(self):
        self.v = 0
        self.lock = threading.Lock()

    def next(self) -> int:
        with self.lock:
            self.v += 1
            return self.v


_worker_nonce = _rand(8)
_obj_seq = _Seq()
_task_seq = _Seq()


def new_object_id() -> bytes:
    """16 bytes: 8-byte worker nonce + 8-byte sequence."""
    return _worker_nonce + struct.pack("<Q", _obj_seq.next())


def new_task_id() -> bytes:
    return (_worker_nonce + struct.pack("<Q", _task_seq.next()))[:TASK_ID_LEN]


def new_actor_id() -> bytes:
    return _rand(ACTOR_ID_LEN)


def new_node_id() -> bytes:
    return _rand(NODE_ID_LEN)


def new_pg_id() -> bytes:
    return _rand(PG_ID_LEN)


def hex_id(b: bytes) -> str:
    return b.hex()
