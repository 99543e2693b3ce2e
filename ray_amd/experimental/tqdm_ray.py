"""Distributed-safe progress bars (reference: python/ray/experimental/
tqdm_ray.py) — prints positioned progress lines from any worker."""
from __future__ import annotations

import sys
import time


class tqdm:
    def __init__(self, iterable=None, desc="", total=None, position=0, **kw):
        self.iterable = iterable
        self.desc = desc
        self.total = total or (len(iterable) if iterable is not None and hasattr(iterable, "__len__") else None)
        self.n = 0
        self._last = 0.0

    def update(self, n=1):
        self.n += n
        now = time.time()
        if now - self._last > 0.25:
            self._last = now
            tot = f"/{self.total}" if self.total else ""
            sys.stderr.write(f"\r{self.desc}: {self.n}{tot}")
            sys.stderr.flush()

    def __iter__(self):
        for x in self.iterable:
            yield x
            self.update(1)
        self.close()

    def close(self):
        sys.stderr.write("\n")

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()
