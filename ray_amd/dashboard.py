"""Dashboard-lite (reference: python/ray/dashboard/ — head + REST
routes + UI). ray_amd serves the same cluster state as a JSON REST API
plus a minimal HTML overview; start with `python -m ray_amd dashboard`
or mount the ASGI app in tests."""
from __future__ import annotations

import json
import time


def build_asgi_app():
    """Plain ASGI app (no framework) serving cluster state."""

    async def app(scope, receive, send):
        if scope["type"] != "http":
            return
        path = scope["path"]
        status, ctype, body = 200, "application/json", b"{}"
        try:
            if path in ("/", "/index.html"):
                ctype, body = "text/html", _INDEX_HTML.encode()
            elif path == "/api/nodes":
                body = _json(_nodes())
            elif path == "/api/actors":
                from ray_amd.util import state as st

                body = _json(st.list_actors(limit=1000))
            elif path == "/api/tasks":
                from ray_amd.util import state as st

                body = _json(st.list_tasks(limit=1000))
            elif path == "/api/cluster_status":
                body = _json(_cluster_status())
            elif path == "/api/jobs":
                body = _json(_jobs())
            elif path == "/api/placement_groups":
                from ray_amd.util import state as st

                body = _json(st.list_placement_groups(limit=1000))
            elif path == "/api/objects":
                from ray_amd.util import state as st

                body = _json(st.list_objects())
            elif path == "/api/debug_state":
                from ray_amd.util.state import node_debug_state

                body = _json(node_debug_state())
            elif path == "/api/serve":
                body = _json(_serve_status())
            elif path == "/api/timeline":
                import ray_amd as ray

                body = _json(ray.timeline() or [])
            elif path == "/api/logs":
                body = _json(_log_files())
            elif path.startswith("/api/logs/"):
                ctype, body = "text/plain", _log_tail(path[len("/api/logs/"):])
            elif path == "/metrics":
                from ray_amd.util import metrics

                ctype, body = "text/plain", metrics.export_text().encode()
            else:
                status, body = 404, b'{"error": "not found"}'
        except Exception as e:  # pragma: no cover
            status, body = 500, json.dumps({"error": str(e)}).encode()
        await send(
            {
                "type": "http.response.start",
                "status": status,
                "headers": [
                    (b"content-type", ctype.encode()),
                    (b"content-length", str(len(body)).encode()),
                ],
            }
        )
        await send({"type": "http.response.body", "body": body})

    return app


def _json(obj) -> bytes:
    return json.dumps(obj, default=str).encode()


def _nodes():
    import ray_amd as ray

    return ray.nodes()


def _jobs():
    """Job manager listing (reference: dashboard/modules/job)."""
    import ray_amd as ray
    from ray_amd.job_submission import (
        JOB_MANAGER_NAME,
        JOB_NAMESPACE,
    )

    try:
        mgr = ray.get_actor(JOB_MANAGER_NAME, namespace=JOB_NAMESPACE)
        return ray.get(mgr.list.remote(), timeout=10)
    except Exception:
        return []


def _serve_status():
    """Serve app/deployment/replica view (reference:
    dashboard/modules/serve)."""
    import ray_amd as ray

    try:
        from ray_amd.serve.api import SERVE_CONTROLLER_NAME, SERVE_NAMESPACE

        ctrl = ray.get_actor(SERVE_CONTROLLER_NAME, namespace=SERVE_NAMESPACE)
        return ray.get(ctrl.status.remote(), timeout=10)
    except Exception:
        return {}


def _log_files():
    from ray_amd._core import runtime as rtmod
    import os

    rt = rtmod.global_runtime()
    d = os.path.join(rt.session_dir, "logs")
    try:
        return sorted(os.listdir(d))
    except OSError:
        return []


def _log_tail(name: str, n: int = 200) -> bytes:
    from ray_amd._core import runtime as rtmod
    import os

    if "/" in name or ".." in name:
        return b"bad log name"
    rt = rtmod.global_runtime()
    p = os.path.join(rt.session_dir, "logs", name)
    try:
        with open(p, "rb") as f:
            return b"\n".join(f.read().splitlines()[-n:])
    except OSError:
        return b"no such log"


def _cluster_status():
    import ray_amd as ray
    from ray_amd.util import state as st

    actors = st.list_actors(limit=10000)
    return {
        "timestamp": time.time(),
        "nodes": len([n for n in ray.nodes() if n["Alive"]]),
        "resources_total": ray.cluster_resources(),
        "resources_available": ray.available_resources(),
        "actors_alive": sum(1 for a in actors if a["state"] == "ALIVE"),
        "actors_total": len(actors),
        "object_store": st.list_objects()[0],
    }


_INDEX_HTML = """<!doctype html>
<html><head><title>ray_amd dashboard</title>
<style>body{font-family:monospace;margin:2em}pre{background:#f4f4f4;padding:1em}</style>
</head><body>
<h2>ray_amd cluster</h2>
<pre id="status">loading…</pre>
<script>
async function refresh(){
  const r = await fetch('/api/cluster_status');
  document.getElementById('status').textContent =
      JSON.stringify(await r.json(), null, 2);
}
refresh(); setInterval(refresh, 2000);
</script>
<div id="tabs"></div>
<pre id="detail"></pre>
<script>
const APIS=["nodes","actors","tasks","jobs","placement_groups","objects",
            "serve","debug_state","logs"];
const tabs=document.getElementById('tabs');
APIS.forEach(a=>{
  const b=document.createElement('button'); b.textContent=a;
  b.onclick=async()=>{const r=await fetch('/api/'+a);
    document.getElementById('detail').textContent=
      JSON.stringify(await r.json(),null,2);};
  tabs.appendChild(b);});
</script>
<p>APIs: /api/{nodes,actors,tasks,jobs,placement_groups,objects,serve,
debug_state,timeline,logs,logs/&lt;file&gt;} /metrics</p>
</body></html>"""


def run_dashboard(host: str = "127.0.0.1", port: int = 8265):
    import uvicorn

    uvicorn.run(build_asgi_app(), host=host, port=port, log_level="warning")
