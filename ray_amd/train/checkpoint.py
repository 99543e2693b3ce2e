"""Checkpoint — directory-based, byte-compatible layout with the
reference (python/ray/train/_checkpoint.py:56): a checkpoint IS a
directory (plus optional metadata.json); from_directory/to_directory/
as_directory have identical semantics so reference-written checkpoints
load here and vice versa.
"""
from __future__ import annotations

import contextlib
import json
import os
import shutil
import tempfile
import uuid
from typing import Any, Dict, Optional

_METADATA_FILE = ".metadata.json"


class Checkpoint:
    def __init__(self, path: str, filesystem=None):
        self.path = os.path.abspath(path)
        self.filesystem = filesystem

    @classmethod
    def from_directory(cls, path: str) -> "Checkpoint":
        return cls(path)

    def to_directory(self, path: Optional[str] = None) -> str:
        dest = path or os.path.join(
            tempfile.gettempdir(), f"ckpt_{uuid.uuid4().hex[:8]}"
        )
        if os.path.abspath(dest) != self.path:
            os.makedirs(dest, exist_ok=True)
            shutil.copytree(self.path, dest, dirs_exist_ok=True)
        return dest

    @contextlib.contextmanager
    def as_directory(self):
        # local filesystem: hand out the path directly (zero-copy)
        yield self.path

    def get_metadata(self) -> Dict[str, Any]:
        p = os.path.join(self.path, _METADATA_FILE)
        if os.path.exists(p):
            with open(p) as f:
                return json.load(f)
        return {}

    def set_metadata(self, metadata: Dict[str, Any]):
        with open(os.path.join(self.path, _METADATA_FILE), "w") as f:
            json.dump(metadata, f)

    def update_metadata(self, metadata: Dict[str, Any]):
        m = self.get_metadata()
        m.update(metadata)
        self.set_metadata(m)

    def __repr__(self):
        return f"Checkpoint(path={self.path})"

    def __reduce__(self):
        return (Checkpoint, (self.path,))
