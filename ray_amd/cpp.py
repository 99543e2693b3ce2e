"""C++ task bodies (reference: cpp/ worker API): user C++ functions in
a shared library run inside ray_amd workers via one extern-C ABI
(csrc/cpp_client/task_api.hpp). The library is dlopen'd lazily per
worker and cached."""
from __future__ import annotations

import ctypes
import struct
from typing import List

_libs = {}


def _load(lib_path: str):
    lib = _libs.get(lib_path)
    if lib is None:
        lib = ctypes.CDLL(lib_path)
        lib.ray_amd_cpp_invoke.restype = ctypes.c_int
        lib.ray_amd_cpp_invoke.argtypes = [
            ctypes.c_char_p, ctypes.c_char_p, ctypes.c_long,
            ctypes.POINTER(ctypes.c_char_p), ctypes.POINTER(ctypes.c_long),
        ]
        lib.ray_amd_cpp_list.restype = ctypes.c_char_p
        _libs[lib_path] = lib
    return lib


def invoke(lib_path: str, fn_name: str, payload: bytes) -> bytes:
    lib = _load(lib_path)
    out = ctypes.c_char_p()
    out_len = ctypes.c_long()
    rc = lib.ray_amd_cpp_invoke(
        fn_name.encode(), payload, len(payload),
        ctypes.byref(out), ctypes.byref(out_len),
    )
    if rc == 1:
        names = (lib.ray_amd_cpp_list() or b"").decode()
        raise ValueError(
            f"C++ task {fn_name!r} not registered in {lib_path} "
            f"(has: {names})"
        )
    if rc != 0:
        raise RuntimeError(f"C++ task {fn_name!r} raised")
    data = ctypes.string_at(out, out_len.value)
    lib.ray_amd_cpp_free(out)
    return data


def list_functions(lib_path: str) -> List[str]:
    names = (_load(lib_path).ray_amd_cpp_list() or b"").decode()
    return [n for n in names.split(",") if n]


def remote_function(lib_path: str, fn_name: str, **options):
    """A remote handle whose .remote(payload: bytes) executes the C++
    function inside a worker."""
    import ray_amd as ray

    @ray.remote(**options)
    def _cpp_task(payload: bytes, _lib=lib_path, _fn=fn_name):
        from ray_amd import cpp as _cpp

        return _cpp.invoke(_lib, _fn, payload)

    class _Handle:
        def remote(self, payload: bytes = b""):
            return _cpp_task.remote(payload)

    return _Handle()


# payload helpers mirroring task_api.hpp
def pack_i64(v: int) -> bytes:
    return struct.pack("<q", v)


def unpack_i64(b: bytes) -> int:
    return struct.unpack("<q", b[:8])[0]


def pack_pair_i64(a: int, b: int) -> bytes:
    return struct.pack("<qq", a, b)


def pack_f64_vec(v) -> bytes:
    import numpy as np

    return np.asarray(v, np.float64).tobytes()


def unpack_f64_vec(b: bytes):
    import numpy as np

    return np.frombuffer(b, np.float64)
