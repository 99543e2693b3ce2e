"""Serialization debugging (reference: python/ray/util/check_serialize.py)."""
from __future__ import annotations

from typing import Any, Set, Tuple


def inspect_serializability(obj: Any, name: str = "obj") -> Tuple[bool, Set]:
    import cloudpickle

    failures = set()
    try:
        cloudpickle.dumps(obj)
        return True, failures
    except Exception as e:
        failures.add((name, str(e)))
        for attr in ("__dict__",):
            d = getattr(obj, attr, None)
            if isinstance(d, dict):
                for k, v in d.items():
                    try:
                        cloudpickle.dumps(v)
                    except Exception as e2:
                        failures.add((f"{name}.{k}", str(e2)))
        return False, failures
