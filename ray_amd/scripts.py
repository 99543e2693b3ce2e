"""`ray_amd` CLI (reference: python/ray/scripts/scripts.py — start :800,
stop :1341, status :2386, timeline :2289, memory :2346,
microbenchmark :2275). Invoke as `python -m ray_amd <cmd>`."""
from __future__ import annotations

import argparse
import json
import os
import signal
import sys
import time


def cmd_start(args):
    from ray_amd._core import node as _node

    if not args.head:
        print("only --head is supported in-round; worker nodes join via "
              "cluster_utils.Cluster", file=sys.stderr)
        return 1
    cluster = _node.start_local_cluster(
        num_cpus=args.num_cpus, num_gpus=args.num_gpus
    )
    with open(os.path.join(cluster.session_dir, "head_pids"), "w") as f:
        json.dump(
            {"gcs": cluster.gcs_proc.pid, "raylet": cluster.raylet_proc.pid}, f
        )
    print(f"started head; session_dir={cluster.session_dir}")
    print(f"connect with: ray_amd.init(address='{cluster.session_dir}')")
    return 0


def cmd_stop(args):
    from ray_amd._core.node import RAY_AMD_TMP

    session = getattr(args, "session", None)
    if not session:
        latest = os.path.join(RAY_AMD_TMP, "latest_session")
        if not os.path.exists(latest):
            print("no running session found")
            return 0
        with open(latest) as f:
            session = f.read().strip()
    pid_file = os.path.join(session, "head_pids")
    if os.path.exists(pid_file):
        with open(pid_file) as f:
            pids = json.load(f)
        for name, pid in pids.items():
            try:
                os.kill(pid, signal.SIGTERM)
                print(f"stopped {name} (pid {pid})")
            except OSError:
                pass
    return 0


def cmd_status(args):
    import ray_amd as ray

    ray.init(address=args.address or "auto")
    print("Nodes:")
    for n in ray.nodes():
        state = "ALIVE" if n["Alive"] else "DEAD"
        print(f"  {n['NodeID'][:12]} {state} total={n['Resources']} "
              f"avail={n['Available']}")
    from ray_amd.util import state as state_api

    actors = state_api.list_actors()
    alive = sum(1 for a in actors if a["state"] == "ALIVE")
    print(f"Actors: {alive} alive / {len(actors)} total")
    st = state_api.list_objects()[0]
    print(f"Object store: {st['num_objects_in_store']} objects, "
          f"{st['store_used_bytes'] / 1e6:.1f} MB used")
    ray.shutdown()
    return 0


def cmd_timeline(args):
    import ray_amd as ray

    ray.init(address=args.address or "auto")
    from ray_amd.util import state as state_api

    tasks = state_api.list_tasks()
    events = []
    for t in tasks:
        events.append(
            {
                "cat": "task",
                "name": t["name"],
                "ph": "X",
                "ts": t["start_time_ms"] * 1000,
                "dur": (t["end_time_ms"] - t["start_time_ms"]) * 1000,
                "pid": t["worker_pid"],
                "tid": t["worker_pid"],
            }
        )
    out = args.output or f"timeline-{time.strftime('%Y%m%d-%H%M%S')}.json"
    with open(out, "w") as f:
        json.dump(events, f)
    print(f"wrote {len(events)} events to {out} (open in chrome://tracing)")
    ray.shutdown()
    return 0


def cmd_memory(args):
    import ray_amd as ray

    ray.init(address=args.address or "auto")
    rt = ray.api._rt.global_runtime()
    st = rt.raylet_call("object_stats", {})
    print(f"shm store: {st['num_objects']} objects, "
          f"{st['used'] / 1e6:.1f}/{st['capacity'] / 1e6:.1f} MB")
    ray.shutdown()
    return 0


def cmd_microbenchmark(args):
    from ray_amd.microbenchmark import main as micro

    micro(args.duration)
    return 0


def main(argv=None):
    p = argparse.ArgumentParser(prog="ray_amd")
    sub = p.add_subparsers(dest="cmd", required=True)

    s = sub.add_parser("start")
    s.add_argument("--head", action="store_true")
    s.add_argument("--num-cpus", type=float, default=None)
    s.add_argument("--num-gpus", type=float, default=None)
    s.set_defaults(fn=cmd_start)

    s = sub.add_parser("stop")
    s.set_defaults(fn=cmd_stop)

    for name, fn in (
        ("status", cmd_status),
        ("timeline", cmd_timeline),
        ("memory", cmd_memory),
    ):
        s = sub.add_parser(name)
        s.add_argument("--address", default=None)
        if name == "timeline":
            s.add_argument("--output", default=None)
        s.set_defaults(fn=fn)

    s = sub.add_parser("dashboard")
    s.add_argument("--address", default=None)
    s.add_argument("--port", type=int, default=8265)
    def _dash(args):
        import ray_amd as ray

        ray.init(address=args.address or "auto")
        from ray_amd.dashboard import run_dashboard

        print(f"dashboard on http://127.0.0.1:{args.port}")
        run_dashboard(port=args.port)
        return 0
    s.set_defaults(fn=_dash)

    s = sub.add_parser("up")
    s.add_argument("config", help="cluster YAML (local provider)")

    def _up(args):
        # reference: `ray up cluster.yaml` (autoscaler/commands.py) —
        # local provider: start a head + worker raylets on this machine
        import yaml

        from ray_amd._core import node as _node

        with open(args.config) as f:
            cfg = yaml.safe_load(f) or {}
        provider = (cfg.get("provider") or {}).get("type", "local")
        if provider != "local":
            print(f"provider {provider!r} needs cloud access; only "
                  "'local' is supported offline", file=sys.stderr)
            return 1
        head = cfg.get("head_node") or {}
        cluster = _node.start_local_cluster(
            num_cpus=head.get("num_cpus"), num_gpus=head.get("num_gpus"),
            resources=head.get("resources"),
        )
        pids = {"gcs": cluster.gcs_proc.pid,
                "raylet": cluster.raylet_proc.pid}
        wn = cfg.get("worker_nodes") or {}
        n_workers = int(wn.get("count", cfg.get("max_workers", 0)))
        for i in range(n_workers):
            res = dict(wn.get("resources") or {})
            res.setdefault("CPU", wn.get("num_cpus", 2))
            res.setdefault("GPU", wn.get("num_gpus", 0))
            res.setdefault("memory", 16 * 2**30)
            proc, addr, node_id = _node.start_raylet(
                cluster.session_dir, cluster.gcs_addr, res,
                node_name=f"worker{i}",
                labels=wn.get("labels"),
            )
            pids[f"worker{i}"] = proc.pid
        with open(os.path.join(cluster.session_dir, "head_pids"),
                  "w") as f:
            json.dump(pids, f)
        name = cfg.get("cluster_name", "local")
        print(f"cluster {name!r} up: head + {n_workers} workers; "
              f"session_dir={cluster.session_dir}")
        print(f"connect with: ray_amd.init(address="
              f"'{cluster.session_dir}')")
        return 0

    s.set_defaults(fn=_up)

    s = sub.add_parser("down")
    s.add_argument("session", nargs="?", default=None,
                   help="session_dir (default: latest)")

    def _down(args):
        return cmd_stop(args)

    s.set_defaults(fn=_down)

    s = sub.add_parser("stack")
    s.add_argument("--address", default=None)

    def _stack(args):
        # reference: `ray stack` (py-spy dump of every worker); here
        # each worker serves its own thread stacks over RPC
        import ray_amd as ray

        ray.init(address=args.address or "auto", ignore_reinit_error=True)
        from ray_amd._core import runtime as _rt

        rt = _rt.global_runtime()

        async def _collect():
            out = []
            for n in ray.nodes():
                if not n["Alive"]:
                    continue
                try:
                    c = await rt._conn(n["Address"])
                    workers = await c.call("list_workers", {})
                except Exception as e:
                    out.append((n["NodeID"], None, f"<unreachable: {e}>"))
                    continue
                for w in workers:
                    try:
                        wc = await rt._conn(w["addr"])
                        d = await asyncio.wait_for(
                            wc.call("dump_stack", {}), 10)
                        out.append((n["NodeID"], w, d))
                    except Exception as e:
                        out.append((n["NodeID"], w, f"<no dump: {e}>"))
            return out

        import asyncio

        for node_id, w, dump in rt._call_sync(_collect()):
            hdr = (f"=== node {node_id[:12]} pid {w['pid']} "
                   f"({w['kind']}) ===" if w else
                   f"=== node {node_id[:12]} ===")
            print(hdr)
            if isinstance(dump, str):
                print(f"  {dump}")
                continue
            for tname, stack in dump["stacks"].items():
                print(f"--- {tname} ---")
                print(stack)
        return 0

    s.set_defaults(fn=_stack)

    s = sub.add_parser("debug")
    s.add_argument("--address", default=None)

    def _debug(args):
        from ray_amd.util.rpdb import cmd_debug

        return cmd_debug(args)

    s.set_defaults(fn=_debug)

    s = sub.add_parser("microbenchmark")
    s.add_argument("--duration", type=float, default=2.0)
    s.set_defaults(fn=cmd_microbenchmark)

    # ray_amd serve run module:app (reference: `serve run`)
    sv = sub.add_parser("serve")
    svsub = sv.add_subparsers(dest="serve_cmd", required=True)
    sr = svsub.add_parser("run")
    sr.add_argument("--address", default=None)
    sr.add_argument("--name", default="default")
    sr.add_argument("--route-prefix", default="/")
    sr.add_argument("--port", type=int, default=8000)
    sr.add_argument("--blocking", action="store_true")
    sr.add_argument("import_path", help="module:app, e.g. my_app:app")

    def _serve_run(args):
        import importlib

        import ray_amd as ray
        from ray_amd import serve as _serve

        ray.init(address=args.address, ignore_reinit_error=True)
        mod_name, _, attr = args.import_path.partition(":")
        sys.path.insert(0, os.getcwd())
        app = getattr(importlib.import_module(mod_name), attr or "app")
        _serve.run(app, name=args.name, route_prefix=args.route_prefix,
                   port=args.port)
        print(f"serving {args.import_path} at "
              f"http://127.0.0.1:{args.port}{args.route_prefix}")
        if args.blocking:
            while True:
                time.sleep(3600)
        return 0

    sr.set_defaults(fn=_serve_run)

    # ray_amd job submit|status|logs|stop|list (reference: `ray job ...`)
    j = sub.add_parser("job")
    jsub = j.add_subparsers(dest="job_cmd", required=True)

    def _client(args):
        from ray_amd.job_submission import JobSubmissionClient

        return JobSubmissionClient(address=args.address)

    js = jsub.add_parser("submit")
    js.add_argument("--address", default=None)
    js.add_argument("--submission-id", default=None)
    js.add_argument("--working-dir", default=None)
    js.add_argument("--no-wait", action="store_true")
    js.add_argument("entrypoint", nargs=argparse.REMAINDER)

    def _submit(args):
        from ray_amd.job_submission import JobStatus

        import shlex

        c = _client(args)
        parts = list(args.entrypoint)
        if parts and parts[0] == "--":
            parts = parts[1:]
        ep = shlex.join(parts)
        renv = {}
        if args.working_dir:
            renv["working_dir"] = args.working_dir
        jid = c.submit_job(entrypoint=ep, runtime_env=renv or None,
                           submission_id=args.submission_id)
        print(jid)
        if args.no_wait:
            return 0
        for chunk in c.tail_job_logs(jid):
            print(chunk, end="")
        return 0 if c.get_job_status(jid) == JobStatus.SUCCEEDED else 1

    js.set_defaults(fn=_submit)

    for jname in ("status", "logs", "stop"):
        js = jsub.add_parser(jname)
        js.add_argument("--address", default=None)
        js.add_argument("submission_id")

        def _mk(jname):
            def run(args):
                c = _client(args)
                if jname == "status":
                    print(c.get_job_status(args.submission_id))
                elif jname == "logs":
                    print(c.get_job_logs(args.submission_id), end="")
                else:
                    print(c.stop_job(args.submission_id))
                return 0

            return run

        js.set_defaults(fn=_mk(jname))

    js = jsub.add_parser("list")
    js.add_argument("--address", default=None)

    def _list(args):
        for j in _client(args).list_jobs():
            print(f"{j.submission_id}  {j.status}  {j.entrypoint[:60]}")
        return 0

    js.set_defaults(fn=_list)

    args = p.parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    sys.exit(main())
