"""Remote pdb for tasks/actors (reference: python/ray/util/rpdb.py —
set_trace() parks the worker on a listening socket, registers the
breakpoint in the GCS, and `ray debug` attaches from the driver
machine).

ray_amd.util.rpdb.set_trace() inside a remote function:
  1. binds a TCP socket on the worker's node (127.0.0.1:ephemeral),
  2. advertises {addr, pid, function, filename, line} under the
     "rpdb" KV namespace,
  3. blocks in accept(); the attached client gets a full Pdb REPL over
     the socket (continue/quit detach and resume the task).

`ray_amd debug` lists active breakpoints and attaches interactively.
"""
from __future__ import annotations

import json
import os
import pdb
import socket
import sys
import time

_NS = "rpdb"


class _RemotePdb(pdb.Pdb):
    """Pdb whose stdin/stdout ride a socket (separate read/write
    files — a single rw file deadlocks on reentrant buffered IO).
    Socket cleanup happens inside continue/quit AFTER set_continue has
    disarmed tracing: a traced close() would re-enter the debugger on
    its own IO."""

    def __init__(self, rfile, wfile, cleanup):
        super().__init__(stdin=rfile, stdout=wfile)
        self.use_rawinput = False
        self.prompt = "(ray-pdb) "
        self._cleanup = cleanup

    def do_continue(self, arg):
        r = super().do_continue(arg)
        self._cleanup()
        return r

    do_c = do_cont = do_continue

    def do_quit(self, arg):
        r = super().do_quit(arg)
        self._cleanup()
        return r

    do_q = do_exit = do_quit

    def do_EOF(self, arg):
        r = super().do_EOF(arg)
        self._cleanup()
        return r


def _kv():
    from ray_amd.experimental import internal_kv

    return internal_kv


def set_trace(breakpoint_uuid: str = None, frame=None):
    """Park this worker on a debugger socket until a client attaches
    (or RAY_AMD_RPDB_TIMEOUT_S elapses, default 300 — the task then
    continues rather than hanging a production job forever)."""
    bp_id = breakpoint_uuid or os.urandom(6).hex()
    srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    host = os.environ.get("RAY_AMD_NODE_IP", "127.0.0.1")
    srv.bind((host, 0))
    srv.listen(1)
    addr = f"{srv.getsockname()[0]}:{srv.getsockname()[1]}"

    caller = frame or sys._getframe(1)
    rec = {
        "addr": addr,
        "pid": os.getpid(),
        "function": caller.f_code.co_name,
        "filename": caller.f_code.co_filename,
        "line": caller.f_lineno,
        "ts": time.time(),
    }
    key = f"bp:{bp_id}".encode()
    try:
        _kv()._internal_kv_put(key, json.dumps(rec).encode(), namespace=_NS)
    except Exception:
        pass  # debugger still usable via printed address
    print(f"[ray_amd rpdb] breakpoint {bp_id} waiting at {addr} "
          f"({rec['function']} {rec['filename']}:{rec['line']}) — "
          f"attach with: ray_amd debug", file=sys.stderr, flush=True)

    timeout = float(os.environ.get("RAY_AMD_RPDB_TIMEOUT_S", "300"))
    srv.settimeout(timeout)
    try:
        conn, _peer = srv.accept()
    except socket.timeout:
        print(f"[ray_amd rpdb] breakpoint {bp_id}: no client within "
              f"{timeout}s — continuing", file=sys.stderr, flush=True)
        return
    finally:
        try:
            _kv()._internal_kv_del(key, namespace=_NS)
        except Exception:
            pass
        srv.close()

    rf = conn.makefile("r")
    wf = conn.makefile("w", buffering=1)

    def _cleanup():
        for c in (rf, wf, conn):
            try:
                c.close()
            except OSError:
                pass

    dbg = _RemotePdb(rf, wf, _cleanup)
    dbg.set_trace(caller)


def list_breakpoints():
    """Active breakpoints as {id: record} from the GCS."""
    kv = _kv()
    out = {}
    for k in kv._internal_kv_list(b"bp:", namespace=_NS):
        v = kv._internal_kv_get(bytes(k), namespace=_NS)
        if v:
            out[bytes(k).decode()[3:]] = json.loads(v)
    return out


def attach(addr: str, stdin=None, stdout=None):
    """Connect to a parked breakpoint and bridge the Pdb REPL to this
    terminal. Returns when the debugger detaches (continue/quit)."""
    stdin = stdin or sys.stdin
    stdout = stdout or sys.stdout
    host, _, port = addr.rpartition(":")
    s = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    s.connect((host, int(port)))
    s.settimeout(0.2)
    buf = b""
    try:
        while True:
            # pump debugger output until it blocks on the prompt
            try:
                data = s.recv(4096)
                if not data:
                    break
                buf += data
                stdout.write(data.decode(errors="replace"))
                stdout.flush()
                continue
            except socket.timeout:
                pass
            if b"(ray-pdb)" in buf:
                buf = b""
                line = stdin.readline()
                if not line:
                    line = "c\n"
                s.sendall(line.encode())
                if line.strip() in ("c", "cont", "continue", "q", "quit"):
                    # drain any final output then detach
                    try:
                        s.settimeout(0.5)
                        tail = s.recv(4096)
                        if tail:
                            stdout.write(tail.decode(errors="replace"))
                    except OSError:
                        pass
                    break
    finally:
        s.close()


def cmd_debug(args):
    """`ray_amd debug` — list parked breakpoints and attach."""
    import ray_amd as ray

    ray.init(address=getattr(args, "address", None) or "auto",
             ignore_reinit_error=True)
    bps = list_breakpoints()
    if not bps:
        print("no active breakpoints")
        return 0
    items = sorted(bps.items(), key=lambda kv: kv[1]["ts"])
    for i, (bid, rec) in enumerate(items):
        print(f"[{i}] {bid} pid={rec['pid']} {rec['function']} "
              f"{rec['filename']}:{rec['line']} @ {rec['addr']}")
    pick = 0
    if len(items) > 1:
        try:
            pick = int(input("attach to breakpoint #: ") or "0")
        except (ValueError, EOFError):
            pick = 0
    print(f"attaching to {items[pick][1]['addr']} "
          "(Ctrl-D or 'c' to continue the task)")
    attach(items[pick][1]["addr"])
    return 0
