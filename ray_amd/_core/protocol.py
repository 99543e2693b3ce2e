"""Framed msgpack RPC over asyncio (unix-domain or TCP sockets).

This is ray_amd's control-plane transport — the role gRPC plays in the
reference (src/ray/rpc/grpc_server.h:94, server_call.h:110). Re-designed
rather than ported: a single length-prefixed msgpack framing with
request/reply correlation and one-way notifies, over asyncio.Protocol.
On one node every channel is a unix socket; cross-node channels are TCP.

Wire format: 4-byte LE frame length, then msgpack array:
  [MSG_REQUEST, seq:int, method:str, payload]
  [MSG_REPLY,   seq:int, payload]
  [MSG_ERROR,   seq:int, err:str]
  [MSG_NOTIFY,  0,       method:str, payload]

Chaos hooks (reference rpc_chaos.h:24): RAY_AMD_TESTING_RPC_FAILURE env
("method:prob") drops matching requests with the given probability.
"""
from __future__ import annotations

import asyncio
import os
import random
import struct
import traceback
from typing import Any, Awaitable, Callable, Dict, Optional

import msgpack

MSG_REQUEST = 0
MSG_REPLY = 1
MSG_ERROR = 2
MSG_NOTIFY = 3

_HDR = struct.Struct("<I")

_chaos_spec = None


def _chaos_should_drop(method: str) -> bool:
    global _chaos_spec
    if _chaos_spec is None:
        spec = os.environ.get("RAY_AMD_TESTING_RPC_FAILURE", "")
        if spec:
            m, _, p = spec.partition(":")
            _chaos_spec = (m, float(p or "0"))
        else:
            _chaos_spec = ("", 0.0)
    m, p = _chaos_spec
    return bool(m) and (m == "*" or m == method) and random.random() < p


def pack(obj: Any) -> bytes:
    return msgpack.packb(obj, use_bin_type=True)


def unpack(b: bytes) -> Any:
    return msgpack.unpackb(b, raw=False, strict_map_key=False)


class RpcError(Exception):
    pass


class ConnectionLost(Exception):
    pass


class _FramedProtocol(asyncio.Protocol):
    """Shared framing for client and server connections."""

    def __init__(self, on_message, on_lost=None):
        self._on_message = on_message
        self._on_lost = on_lost
        self._buf = bytearray()
        self.transport: Optional[asyncio.Transport] = None

    def connection_made(self, transport):
        self.transport = transport
        try:
            sock = transport.get_extra_info("socket")
            if sock is not None and sock.family == 2:  # AF_INET
                import socket as _s

                sock.setsockopt(_s.IPPROTO_TCP, _s.TCP_NODELAY, 1)
        except Exception:
            pass

    def data_received(self, data: bytes):
        buf = self._buf
        buf += data
        view_start = 0
        n = len(buf)
        while n - view_start >= 4:
            (length,) = _HDR.unpack_from(buf, view_start)
            if n - view_start - 4 < length:
                break
            frame = bytes(buf[view_start + 4 : view_start + 4 + length])
            view_start += 4 + length
            try:
                self._on_message(unpack(frame), self)
            except Exception:
                traceback.print_exc()
        if view_start:
            del buf[:view_start]

    def connection_lost(self, exc):
        if self._on_lost is not None:
            self._on_lost(self, exc)

    def send(self, obj: Any):
        data = pack(obj)
        self.transport.write(_HDR.pack(len(data)) + data)


Handler = Callable[..., Any]


class RpcServer:
    """Asyncio RPC server. Handlers are ``async def h(conn, payload)`` or
    plain functions; return value becomes the reply payload."""

    def __init__(self):
        self._handlers: Dict[str, Handler] = {}
        self._server: Optional[asyncio.AbstractServer] = None
        self._conns = set()
        self.on_conn_lost: Optional[Callable] = None
        # per-handler latency stats (reference: instrumented_io_context /
        # event_stats.cc per-handler counts): method -> [count, total_s,
        # max_s]. Sync handlers measure inline; async handlers measure
        # the awaited span.
        self.handler_stats: Dict[str, list] = {}

    def _record(self, method: str, dt: float):
        st = self.handler_stats.get(method)
        if st is None:
            st = self.handler_stats[method] = [0, 0.0, 0.0]
        st[0] += 1
        st[1] += dt
        if dt > st[2]:
            st[2] = dt

    def stats_table(self) -> Dict[str, dict]:
        return {
            m: {"count": c, "total_s": round(t, 6), "mean_us":
                round(t / c * 1e6, 1) if c else 0.0,
                "max_us": round(mx * 1e6, 1)}
            for m, (c, t, mx) in sorted(self.handler_stats.items())
        }

    def route(self, method: str, handler: Handler):
        self._handlers[method] = handler

    async def start_unix(self, path: str):
        loop = asyncio.get_running_loop()
        os.makedirs(os.path.dirname(path), exist_ok=True)
        self._server = await loop.create_unix_server(self._make_proto, path)

    async def start_tcp(self, host: str, port: int) -> int:
        loop = asyncio.get_running_loop()
        self._server = await loop.create_server(self._make_proto, host, port)
        return self._server.sockets[0].getsockname()[1]

    def _make_proto(self):
        p = _FramedProtocol(self._dispatch, self._lost)
        self._conns.add(p)
        return p

    def _lost(self, proto, exc):
        self._conns.discard(proto)
        if self.on_conn_lost is not None:
            try:
                self.on_conn_lost(proto, exc)
            except Exception:
                traceback.print_exc()

    def _dispatch(self, msg, proto):
        import time as _time

        mtype = msg[0]
        if mtype == MSG_REQUEST:
            _, seq, method, payload = msg
            if _chaos_should_drop(method):
                return
            h = self._handlers.get(method)
            if h is None:
                proto.send([MSG_ERROR, seq, f"no such method: {method}"])
                return
            coro_or_val = None
            t0 = _time.perf_counter()
            try:
                coro_or_val = h(proto, payload)
            except Exception:
                self._record(method, _time.perf_counter() - t0)
                proto.send([MSG_ERROR, seq, traceback.format_exc()])
                return
            if asyncio.iscoroutine(coro_or_val):
                task = asyncio.ensure_future(coro_or_val)

                def _done(t, seq=seq, proto=proto, method=method, t0=t0):
                    self._record(method, _time.perf_counter() - t0)
                    if t.cancelled():
                        return
                    e = t.exception()
                    if e is not None:
                        proto.send(
                            [MSG_ERROR, seq, "".join(traceback.format_exception(e))]
                        )
                    elif proto.transport and not proto.transport.is_closing():
                        proto.send([MSG_REPLY, seq, t.result()])

                task.add_done_callback(_done)
            else:
                self._record(method, _time.perf_counter() - t0)
                proto.send([MSG_REPLY, seq, coro_or_val])
        elif mtype == MSG_NOTIFY:
            _, _, method, payload = msg
            h = self._handlers.get(method)
            if h is not None:
                t0 = _time.perf_counter()
                r = h(proto, payload)
                self._record(method, _time.perf_counter() - t0)
                if asyncio.iscoroutine(r):
                    asyncio.ensure_future(r)

    async def close(self):
        if self._server is not None:
            self._server.close()
            await self._server.wait_closed()
        for c in list(self._conns):
            try:
                c.transport.close()
            except Exception:
                pass


async def bind_server(server: "RpcServer", session_dir: str, name: str) -> str:
    """Bind a daemon server on the configured transport. With
    RAY_AMD_NODE_IP set the server listens on TCP at that address
    (multi-machine mode; reference: network-addressed raylet startup,
    _private/node.py:1422); otherwise a unix socket under the session
    dir (single-machine fast path)."""
    node_ip = os.environ.get("RAY_AMD_NODE_IP")
    if node_ip:
        port = await server.start_tcp(node_ip, 0)
        return f"tcp:{node_ip}:{port}"
    sock = os.path.join(session_dir, "sock", name)
    await server.start_unix(sock)
    return "unix:" + sock


class RpcClient:
    """Asyncio RPC client with auto seq correlation. Not thread-safe;
    use from the owning event loop."""

    def __init__(self):
        self._proto: Optional[_FramedProtocol] = None
        self._seq = 0
        self._pending: Dict[int, asyncio.Future] = {}
        self._closed = False
        self.addr = None

    @property
    def connected(self) -> bool:
        return self._proto is not None and not self._closed

    async def connect(self, addr: str, retries: int = 40, delay: float = 0.05):
        """addr: 'unix:/path' or 'tcp:host:port'."""
        self.addr = addr
        loop = asyncio.get_running_loop()
        last = None
        for _ in range(retries):
            try:
                if addr.startswith("unix:"):
                    _, proto = await loop.create_unix_connection(
                        self._make_proto, addr[5:]
                    )
                else:
                    _, host, port = addr.split(":")
                    _, proto = await loop.create_connection(
                        self._make_proto, host, int(port)
                    )
                self._proto = proto
                return
            except (ConnectionRefusedError, FileNotFoundError, OSError) as e:
                last = e
                await asyncio.sleep(delay)
        raise ConnectionError(f"cannot connect to {addr}: {last}")

    def _make_proto(self):
        return _FramedProtocol(self._on_message, self._on_lost)

    def _on_message(self, msg, proto):
        mtype = msg[0]
        if mtype == MSG_REPLY:
            fut = self._pending.pop(msg[1], None)
            if fut is not None and not fut.done():
                fut.set_result(msg[2])
        elif mtype == MSG_ERROR:
            fut = self._pending.pop(msg[1], None)
            if fut is not None and not fut.done():
                fut.set_exception(RpcError(msg[2]))
        elif mtype == MSG_NOTIFY:
            self.on_notify(msg[2], msg[3])

    def on_notify(self, method, payload):  # overridable
        pass

    def _on_lost(self, proto, exc):
        self._closed = True
        err = ConnectionLost(f"connection to {self.addr} lost: {exc}")
        for fut in self._pending.values():
            if not fut.done():
                fut.set_exception(err)
        self._pending.clear()

    async def call(self, method: str, payload: Any) -> Any:
        if self._proto is None or self._closed:
            raise ConnectionLost(f"not connected to {self.addr}")
        self._seq += 1
        seq = self._seq
        fut = asyncio.get_running_loop().create_future()
        self._pending[seq] = fut
        self._proto.send([MSG_REQUEST, seq, method, payload])
        return await fut

    def notify(self, method: str, payload: Any):
        if self._proto is None or self._closed:
            return
        self._proto.send([MSG_NOTIFY, 0, method, payload])

    def close(self):
        self._closed = True
        if self._proto is not None and self._proto.transport is not None:
            self._proto.transport.close()
