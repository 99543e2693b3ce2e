"""Remote debugger tests (reference: ray/util/rpdb.py + `ray debug`).

A task parks on set_trace(); the test discovers the breakpoint via the
GCS registry, attaches over TCP, inspects a local, and continues the
task.
"""
import json
import socket
import threading
import time

import ray_amd as ray


def _read_until(sock, token: bytes, timeout=20.0) -> bytes:
    sock.settimeout(timeout)
    buf = b""
    deadline = time.time() + timeout
    while token not in buf and time.time() < deadline:
        try:
            data = sock.recv(4096)
        except socket.timeout:
            break
        if not data:
            break
        buf += data
    return buf


def test_set_trace_attach_and_continue(ray_start_regular):
    from ray_amd.experimental import internal_kv as kv
    from ray_amd.util import rpdb

    @ray.remote
    def buggy():
        secret = 41  # noqa: F841 — inspected through the debugger
        rpdb.set_trace()
        return "resumed"

    ref = buggy.remote()

    # discover the advertised breakpoint
    deadline = time.time() + 30
    bps = {}
    while not bps and time.time() < deadline:
        bps = rpdb.list_breakpoints()
        time.sleep(0.1)
    assert bps, "breakpoint never registered"
    rec = next(iter(bps.values()))
    assert rec["function"] == "buggy"

    host, _, port = rec["addr"].rpartition(":")
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.connect((host, int(port)))
    _read_until(s, b"(ray-pdb)")
    s.sendall(b"p secret + 1\n")
    out = _read_until(s, b"(ray-pdb)")
    assert b"42" in out, out
    s.sendall(b"c\n")
    s.close()

    assert ray.get(ref, timeout=30) == "resumed"
    # registry cleaned up
    deadline = time.time() + 10
    while rpdb.list_breakpoints() and time.time() < deadline:
        time.sleep(0.1)
    assert not rpdb.list_breakpoints()


def test_set_trace_times_out_without_client(ray_start_regular,
                                            monkeypatch):
    @ray.remote
    def parked():
        import os

        os.environ["RAY_AMD_RPDB_TIMEOUT_S"] = "1.0"
        from ray_amd.util import rpdb as r

        r.set_trace()
        return "continued"

    assert ray.get(parked.remote(), timeout=60) == "continued"


def test_attach_helper_bridges_repl(ray_start_regular):
    """attach() pumps the Pdb prompt to the provided streams."""
    import io

    from ray_amd.util import rpdb

    @ray.remote
    def task():
        rpdb.set_trace()
        return "ok"

    ref = task.remote()
    deadline = time.time() + 30
    bps = {}
    while not bps and time.time() < deadline:
        bps = rpdb.list_breakpoints()
        time.sleep(0.1)
    rec = next(iter(bps.values()))

    out = io.StringIO()
    rpdb.attach(rec["addr"], stdin=io.StringIO("c\n"), stdout=out)
    assert "(ray-pdb)" in out.getvalue()
    assert ray.get(ref, timeout=30) == "ok"


def test_dump_stack_rpc(ray_start_regular):
    """`ray_amd stack` plumbing: raylet lists workers, each worker
    serves its thread stacks over RPC (reference: `ray stack`)."""
    import asyncio

    @ray.remote
    class Sleeper:
        def spin(self):
            time.sleep(0.5)
            return "ok"

    a = Sleeper.remote()
    assert ray.get(a.spin.remote(), timeout=30) == "ok"  # worker is up
    ref = a.spin.remote()
    rt = ray.api._rt.global_runtime()

    async def collect():
        workers = await rt.raylet.call("list_workers", {})
        assert workers, "no workers listed"
        dumps = []
        for w in workers:
            c = await rt._conn(w["addr"])
            dumps.append(await asyncio.wait_for(c.call("dump_stack", {}), 10))
        return dumps

    dumps = rt._call_sync(collect())
    assert all("stacks" in d and d["stacks"] for d in dumps)
    joined = "\n".join("".join(d["stacks"].values()) for d in dumps)
    assert "worker" in joined or "run" in joined  # real frames present
    assert ray.get(ref, timeout=30) == "ok"
