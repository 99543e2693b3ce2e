"""Internal KV shim over the GCS (reference: python/ray/experimental/
internal_kv.py)."""
from __future__ import annotations


def _rt():
    from ray_amd._core import runtime as r

    return r.global_runtime()


def _internal_kv_put(key: bytes, value: bytes, overwrite=True, namespace=None) -> bool:
    return _rt().gcs_call(
        "kv_put",
        {"ns": namespace or "", "key": bytes(key), "value": bytes(value),
         "overwrite": overwrite},
    )


def _internal_kv_get(key: bytes, namespace=None):
    v = _rt().gcs_call("kv_get", {"ns": namespace or "", "key": bytes(key)})
    return bytes(v) if v is not None else None


def _internal_kv_del(key: bytes, namespace=None) -> bool:
    return _rt().gcs_call("kv_del", {"ns": namespace or "", "key": bytes(key)})


def _internal_kv_exists(key: bytes, namespace=None) -> bool:
    return _rt().gcs_call("kv_exists", {"ns": namespace or "", "key": bytes(key)})


def _internal_kv_list(prefix: bytes, namespace=None):
    return [bytes(k) for k in _rt().gcs_call(
        "kv_keys", {"ns": namespace or "", "prefix": bytes(prefix)}
    )]


def _internal_kv_initialized() -> bool:
    from ray_amd._core import runtime as r

    return r.is_initialized()
