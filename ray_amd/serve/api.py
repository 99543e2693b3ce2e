"""Serve public API + controller/replica/router/proxy internals.

Reference counterparts: serve/api.py:522 (@deployment), :952 (run),
serve/handle.py (DeploymentHandle), _private/controller.py:133
(ServeController), _private/replica.py:3249 (ReplicaActor),
_private/proxy.py (HTTP proxy), request_router/pow_2_router.py:27
(power-of-two-choices), batching.py (serve.batch), autoscaling_state.py
(request-rate autoscaling).
"""
from __future__ import annotations

import asyncio
import random
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

# starlette State instances (FastAPI app.state) break pickle: their
# __getattr__ recurses during reconstruction before _state exists.
# Register a proper reducer so serve deployments with FastAPI ingress
# apps serialize (the reference ships its own patched cloudpickle).
try:
    from starlette.datastructures import State as _StarletteState

    def _rebuild_starlette_state(d):
        return _StarletteState(d)

    def _reduce_starlette_state(self):
        return (_rebuild_starlette_state, (dict(self._state),))

    _StarletteState.__reduce__ = _reduce_starlette_state
except ImportError:
    pass

SERVE_CONTROLLER_NAME = "SERVE_CONTROLLER_ACTOR"
SERVE_PROXY_NAME = "SERVE_PROXY_ACTOR"
SERVE_NAMESPACE = "serve"


def _ray():
    import ray_amd as ray

    return ray


# --------------------------------------------------------------------------
# Deployment declaration
# --------------------------------------------------------------------------


@dataclass
class AutoscalingConfig:
    min_replicas: int = 1
    max_replicas: int = 4
    target_ongoing_requests: float = 2.0
    upscale_delay_s: float = 3.0
    downscale_delay_s: float = 30.0


class Deployment:
    def __init__(self, func_or_class, name: str, *, num_replicas=1,
                 max_ongoing_requests=100, ray_actor_options=None,
                 user_config=None, autoscaling_config=None,
                 health_check_period_s=10.0,
                 placement_group_bundles=None,
                 placement_group_strategy="PACK", **kwargs):
        self.func_or_class = func_or_class
        self.name = name
        self.num_replicas = num_replicas
        self.max_ongoing_requests = max_ongoing_requests
        self.ray_actor_options = ray_actor_options or {}
        self.user_config = user_config
        # gang placement per replica (reference: serve deployment
        # placement_group_bundles/placement_group_strategy): each
        # replica reserves its bundles all-or-nothing and runs in
        # bundle 0
        self.placement_group_bundles = placement_group_bundles
        self.placement_group_strategy = placement_group_strategy
        if isinstance(autoscaling_config, dict):
            autoscaling_config = AutoscalingConfig(**autoscaling_config)
        self.autoscaling_config = autoscaling_config

    def options(self, **kwargs) -> "Deployment":
        merged = dict(
            num_replicas=self.num_replicas,
            max_ongoing_requests=self.max_ongoing_requests,
            ray_actor_options=self.ray_actor_options,
            user_config=self.user_config,
            autoscaling_config=self.autoscaling_config,
            placement_group_bundles=self.placement_group_bundles,
            placement_group_strategy=self.placement_group_strategy,
        )
        name = kwargs.pop("name", self.name)
        merged.update(kwargs)
        return Deployment(self.func_or_class, name, **merged)

    def bind(self, *args, **kwargs) -> "Application":
        return Application(DeploymentNode(self, args, kwargs))

    def __call__(self, *a, **k):
        raise RuntimeError(
            "Deployments cannot be called directly; use .bind() and serve.run"
        )


class DeploymentNode:
    def __init__(self, deployment: Deployment, args, kwargs):
        self.deployment = deployment
        self.args = args
        self.kwargs = kwargs


class Application:
    def __init__(self, root: DeploymentNode):
        self.root = root

    def _collect(self) -> List[DeploymentNode]:
        """Topological order, children first."""
        seen: List[DeploymentNode] = []

        def visit(node: DeploymentNode):
            for a in list(node.args) + list(node.kwargs.values()):
                if isinstance(a, Application):
                    visit(a.root)
                elif isinstance(a, DeploymentNode):
                    visit(a)
            if node not in seen:
                seen.append(node)

        visit(self.root)
        return seen


def deployment(_func_or_class=None, *, name=None, num_replicas=1,
               max_ongoing_requests=100, ray_actor_options=None,
               user_config=None, autoscaling_config=None,
               placement_group_bundles=None,
               placement_group_strategy="PACK", **kwargs):
    """@serve.deployment decorator (reference: serve/api.py:522)."""

    def make(fc):
        n = name or getattr(fc, "__name__", "deployment")
        if isinstance(num_replicas, str) and num_replicas == "auto":
            asc = autoscaling_config or AutoscalingConfig()
            nr = asc.min_replicas
        else:
            asc = autoscaling_config
            nr = num_replicas
        return Deployment(
            fc, n, num_replicas=nr, max_ongoing_requests=max_ongoing_requests,
            ray_actor_options=ray_actor_options, user_config=user_config,
            autoscaling_config=asc,
            placement_group_bundles=placement_group_bundles,
            placement_group_strategy=placement_group_strategy,
        )

    if _func_or_class is not None:
        return make(_func_or_class)
    return make


def ingress(asgi_app):
    """@serve.ingress(fastapi_app): attach an ASGI app to the class."""

    def dec(cls):
        cls.__serve_asgi_app__ = asgi_app
        return cls

    return dec


# --------------------------------------------------------------------------
# Replica
# --------------------------------------------------------------------------


class ReplicaActor:
    def __init__(self, cls_or_fn, init_args, init_kwargs, user_config):
        import inspect

        self._ongoing = 0
        self._total = 0
        self._asgi_client = None
        if inspect.isclass(cls_or_fn):
            self._callable = cls_or_fn(*init_args, **init_kwargs)
            asgi = getattr(cls_or_fn, "__serve_asgi_app__", None)
            if asgi is not None:
                import httpx

                self._asgi_app = asgi
                self._asgi_client = httpx.AsyncClient(
                    transport=httpx.ASGITransport(app=asgi),
                    base_url="http://serve",
                )
                # let FastAPI-style ingress classes access self via app state
                try:
                    asgi.state.serve_instance = self._callable
                except Exception:
                    pass
        else:
            self._callable = cls_or_fn
        if user_config is not None and hasattr(self._callable, "reconfigure"):
            self._callable.reconfigure(user_config)

    async def handle_request(self, method: str, args, kwargs):
        self._ongoing += 1
        self._total += 1
        try:
            target = (
                self._callable
                if method == "__call__" and not hasattr(self._callable, "__call__")
                else getattr(self._callable, method, None)
            )
            if target is None and method == "__call__":
                target = self._callable
            if asyncio.iscoroutinefunction(target):
                return await target(*args, **kwargs)
            # sync user code runs off-loop so it may block on .result()
            loop = asyncio.get_running_loop()
            r = await loop.run_in_executor(
                None, lambda: target(*args, **kwargs)
            )
            import inspect as _insp

            # NB: asyncio.iscoroutine() is True for plain generators
            # (legacy coroutines) — generators must pass through intact
            # for streaming responses.
            if _insp.iscoroutine(r) and not _insp.isgenerator(r):
                r = await r
            return r
        finally:
            self._ongoing -= 1

    async def handle_http(self, method: str, path: str, query: str,
                          headers: dict, body: bytes):
        self._ongoing += 1
        self._total += 1
        try:
            if self._asgi_client is not None:
                url = path + (f"?{query}" if query else "")
                resp = await self._asgi_client.request(
                    method, url, headers=headers, content=body
                )
                return resp.status_code, dict(resp.headers), resp.content
            req = SimpleRequest(method, path, query, headers, body)
            r = self._callable(req) if callable(self._callable) else None
            if asyncio.iscoroutine(r):
                r = await r
            return _encode_http_result(r)
        finally:
            self._ongoing -= 1

    def is_http_streaming(self) -> bool:
        """True when the deployment's __call__ is a (sync or async)
        generator function — the proxy then uses handle_http_stream and
        chunked transfer (reference: _private/proxy.py streaming)."""
        import inspect

        if self._asgi_client is not None:
            return False
        target = getattr(self._callable, "__call__", self._callable)
        return inspect.isgeneratorfunction(
            target
        ) or inspect.isasyncgenfunction(target)

    async def handle_http_stream(self, method: str, path: str, query: str,
                                 headers: dict, body: bytes):
        """Async generator: first item is (status, headers), the rest are
        body chunks — driven by the streaming-generator task path."""
        import inspect

        self._ongoing += 1
        self._total += 1
        try:
            req = SimpleRequest(method, path, query, headers, body)
            r = self._callable(req)
            yield (200, {"content-type": "text/plain; charset=utf-8"})
            if inspect.isasyncgen(r):
                async for chunk in r:
                    yield _chunk_bytes(chunk)
            else:
                it = iter(r)
                loop = asyncio.get_running_loop()
                sentinel = object()
                while True:
                    chunk = await loop.run_in_executor(
                        None, next, it, sentinel
                    )
                    if chunk is sentinel:
                        break
                    yield _chunk_bytes(chunk)
        finally:
            self._ongoing -= 1

    async def handle_grpc(self, method: str, data: bytes) -> bytes:
        """gRPC ingress entry (reference: replica gRPC user-method
        dispatch): the named method (or __call__) gets a GrpcRequest;
        bytes results pass through, str encodes, other values JSON."""
        self._ongoing += 1
        self._total += 1
        try:
            req = GrpcRequest(method, data)
            target = getattr(self._callable, method, None)
            if target is None:
                target = self._callable
            r = target(req)
            if asyncio.iscoroutine(r):
                r = await r
            if isinstance(r, (bytes, bytearray)):
                return bytes(r)
            if isinstance(r, str):
                return r.encode()
            import json

            return json.dumps(r, default=str).encode()
        finally:
            self._ongoing -= 1

    def get_stats(self):
        return {"ongoing": self._ongoing, "total": self._total}

    def check_health(self):
        if hasattr(self._callable, "check_health"):
            self._callable.check_health()
        return True


class SimpleRequest:
    """Minimal starlette-Request-compatible object for plain ingress."""

    def __init__(self, method, path, query, headers, body):
        self.method = method
        self.path = path
        self.headers = headers or {}
        self._body = body
        from urllib.parse import parse_qsl

        self.query_params = dict(parse_qsl(query or ""))

    async def body(self) -> bytes:
        return self._body

    async def json(self):
        import json

        return json.loads(self._body or b"null")


def _chunk_bytes(chunk) -> bytes:
    import json

    if isinstance(chunk, (bytes, bytearray)):
        return bytes(chunk)
    if isinstance(chunk, str):
        return chunk.encode()
    return (json.dumps(chunk, default=str) + "\n").encode()


def _encode_http_result(r):
    import json

    if isinstance(r, (bytes, bytearray)):
        return 200, {"content-type": "application/octet-stream"}, bytes(r)
    if isinstance(r, str):
        return 200, {"content-type": "text/plain"}, r.encode()
    return (
        200,
        {"content-type": "application/json"},
        json.dumps(r, default=str).encode(),
    )


# --------------------------------------------------------------------------
# Controller
# --------------------------------------------------------------------------


class ServeController:
    """Reconciles deployments -> replica actors; serves routing tables."""

    def __init__(self):
        # app -> {deployment_name: {"spec":..., "replicas": [handles]}}
        self.apps: Dict[str, dict] = {}
        self.routes: Dict[str, str] = {}  # route_prefix -> app
        self.version = 0
        import threading

        # reference: ServeController.run_control_loop — periodic
        # autoscaling passes + proxy reconciliation (late-joining
        # nodes get their per-node HTTP proxy without a redeploy)
        self._stop_loop = threading.Event()
        threading.Thread(target=self._control_loop, daemon=True,
                         name="serve-control-loop").start()

    def _control_loop(self):
        while not self._stop_loop.wait(3.0):
            try:
                self.autoscale_once()
            except Exception:
                pass
            try:
                self._reconcile_proxies()
            except Exception:
                pass

    def _reconcile_proxies(self):
        if not self.routes:
            return
        ray = _ray()
        try:  # only when HTTP was enabled (head proxy exists)
            ray.get_actor(SERVE_PROXY_NAME, namespace=SERVE_NAMESPACE)
        except ValueError:
            return
        alive = [n for n in ray.nodes() if n["Alive"]]
        if len(alive) > 1:
            _ensure_node_proxies(_http_port)

    def deploy_application(self, name: str, route_prefix: str,
                           specs: List[dict], ingress_name: str):
        ray = _ray()
        old = self.apps.pop(name, None)
        if old:
            self._teardown(old)
        app = {"deployments": {}, "ingress": ingress_name}
        for spec in specs:
            replicas = [
                self._start_replica(spec) for _ in range(spec["num_replicas"])
            ]
            app["deployments"][spec["name"]] = {
                "spec": spec,
                "replicas": replicas,
                "last_scale": time.time(),
            }
        self.apps[name] = app
        if route_prefix:
            self.routes[route_prefix] = name
        self.version += 1
        return True

    def _start_replica(self, spec: dict):
        ray = _ray()
        import cloudpickle

        cls_or_fn, init_args, init_kwargs = cloudpickle.loads(spec["payload"])
        opts = dict(spec.get("ray_actor_options") or {})
        opts.setdefault("num_cpus", 0.1)
        opts["max_concurrency"] = max(16, spec.get("max_ongoing_requests", 100))
        if spec.get("pg_bundles"):
            # gang placement (reference: deployment
            # placement_group_bundles): reserve this replica's bundles
            # all-or-nothing, run the replica actor in bundle 0
            from ray_amd.util.placement_group import placement_group
            from ray_amd.util.scheduling_strategies import (
                PlacementGroupSchedulingStrategy,
            )

            pg = placement_group(spec["pg_bundles"],
                                 strategy=spec.get("pg_strategy", "PACK"))
            ray.get(pg.ready(), timeout=120)
            self._replica_pgs = getattr(self, "_replica_pgs", {})
            opts["scheduling_strategy"] = PlacementGroupSchedulingStrategy(
                placement_group=pg, placement_group_bundle_index=0)
            RA = ray.remote(ReplicaActor)
            replica = RA.options(**opts).remote(
                cls_or_fn, init_args, init_kwargs, spec.get("user_config"))
            self._replica_pgs[replica] = pg
            return replica
        # Deployment scheduler (reference: deployment_scheduler.py
        # spread): round-robin replicas across alive nodes via soft
        # node affinity so one node's failure doesn't take every
        # replica down.
        if "scheduling_strategy" not in opts:
            try:
                from ray_amd.util.scheduling_strategies import (
                    NodeAffinitySchedulingStrategy,
                )

                alive = [n for n in ray.nodes() if n["Alive"]]
                if len(alive) > 1:
                    self._rr = getattr(self, "_rr", 0) + 1
                    target = alive[self._rr % len(alive)]["NodeID"]
                    opts["scheduling_strategy"] = (
                        NodeAffinitySchedulingStrategy(target, soft=True)
                    )
            except Exception:
                pass
        RA = ray.remote(ReplicaActor)
        return RA.options(**opts).remote(
            cls_or_fn, init_args, init_kwargs, spec.get("user_config")
        )

    def _release_replica_pg(self, r):
        pg = getattr(self, "_replica_pgs", {}).pop(r, None)
        if pg is not None:
            try:
                from ray_amd.util.placement_group import (
                    remove_placement_group,
                )

                remove_placement_group(pg)
            except Exception:
                pass

    def _teardown(self, app: dict):
        ray = _ray()
        for d in app["deployments"].values():
            for r in d["replicas"]:
                try:
                    ray.kill(r)
                except Exception:
                    pass
                self._release_replica_pg(r)

    def get_routing(self, app_name: str, deployment: Optional[str] = None):
        app = self.apps.get(app_name)
        if app is None:
            return None
        dname = deployment or app["ingress"]
        d = app["deployments"].get(dname)
        if d is None:
            return None
        return {"replicas": d["replicas"], "version": self.version,
                "deployment": dname}

    async def listen_for_change(self, app_name: str, deployment,
                                known_version: int, timeout_s: float = 30.0):
        """Long-poll (reference: _private/long_poll.py LongPollHost):
        returns fresh routing once the config version moves past
        `known_version`, or None at timeout (unchanged)."""
        import time as _t

        deadline = _t.monotonic() + timeout_s
        while _t.monotonic() < deadline:
            if self.version != known_version:
                return self.get_routing(app_name, deployment)
            await asyncio.sleep(0.1)
        return None

    def resolve_route(self, path: str):
        best = None
        for prefix, app in self.routes.items():
            if path.startswith(prefix.rstrip("/")) or prefix == "/":
                if best is None or len(prefix) > len(best[0]):
                    best = (prefix, app)
        return best

    def list_routes(self):
        return dict(self.routes)

    def delete_application(self, name: str):
        app = self.apps.pop(name, None)
        if app:
            self._teardown(app)
        self.routes = {k: v for k, v in self.routes.items() if v != name}
        self.version += 1
        return app is not None

    def status(self):
        out = {}
        ray = _ray()
        for name, app in self.apps.items():
            deps = {}
            for dname, d in app["deployments"].items():
                deps[dname] = {
                    "status": "HEALTHY",
                    "replica_states": {"RUNNING": len(d["replicas"])},
                }
            out[name] = {"status": "RUNNING", "deployments": deps}
        return out

    def autoscale_once(self):
        """One reconciliation pass of request-based autoscaling."""
        ray = _ray()
        for app in self.apps.values():
            for d in app["deployments"].values():
                spec = d["spec"]
                asc = spec.get("autoscaling")
                if not asc:
                    continue
                stats = ray.get(
                    [r.get_stats.remote() for r in d["replicas"]], timeout=30
                )
                ongoing = sum(s["ongoing"] for s in stats)
                n = len(d["replicas"])
                target = asc["target_ongoing_requests"]
                desired = max(
                    asc["min_replicas"],
                    min(asc["max_replicas"],
                        int((ongoing + target - 1) // target) or asc["min_replicas"]),
                )
                now = time.time()
                if desired > n:
                    for _ in range(desired - n):
                        d["replicas"].append(self._start_replica(spec))
                    d["last_scale"] = now
                    self.version += 1
                elif desired < n and now - d["last_scale"] > asc.get(
                    "downscale_delay_s", 30.0
                ):
                    # graceful drain (reference: replica graceful
                    # shutdown): pull the victims out of routing first
                    # (version bump -> long-poll push), kill them only
                    # once their in-flight requests hit zero
                    victims = d["replicas"][desired:]
                    d["replicas"] = d["replicas"][:desired]
                    d.setdefault("draining", []).extend(victims)
                    d["last_scale"] = now
                    self.version += 1
                self._reap_drained(d)
        return self.version

    def _reap_drained(self, d):
        ray = _ray()
        still = []
        for r in d.get("draining", []):
            try:
                ongoing = ray.get(r.get_stats.remote(),
                                  timeout=10)["ongoing"]
            except Exception:
                continue  # already dead
            if ongoing <= 0:
                try:
                    ray.kill(r)
                except Exception:
                    pass
                self._release_replica_pg(r)
            else:
                still.append(r)
        d["draining"] = still

    def ping(self):
        return "pong"


# --------------------------------------------------------------------------
# Handle + router (power of two choices)
# --------------------------------------------------------------------------


class DeploymentResponse:
    def __init__(self, ref):
        self._ref = ref

    def result(self, timeout_s: Optional[float] = None):
        return _ray().get(self._ref, timeout=timeout_s)

    def __await__(self):
        from ray_amd._core import runtime as _rt

        rt = _rt.global_runtime()

        async def _get():
            return (await rt.get_async([self._ref]))[0]

        return _get().__await__()

    @property
    def object_ref(self):
        return self._ref


class DeploymentResponseGenerator:
    """Streaming response: iterates the replica's generator output
    (reference: handle.options(stream=True) -> generator of results)."""

    def __init__(self, ref_gen):
        self._gen = ref_gen

    def __iter__(self):
        return self

    def __next__(self):
        ref = next(self._gen)
        return _ray().get(ref)


class _RouterState:
    """Routing state shared by every handle clone of one deployment
    (reference: one LongPollClient per router, not per handle)."""

    def __init__(self):
        self.replicas: List = []
        self.version = -1
        self.counts: Dict[int, int] = {}
        self.watch = None


_router_states: Dict[tuple, _RouterState] = {}


class DeploymentHandle:
    def __init__(self, app_name: str, deployment_name: Optional[str] = None,
                 method_name: str = "__call__", stream: bool = False):
        self.app_name = app_name
        self.deployment_name = deployment_name
        self.method_name = method_name
        self._stream = stream
        self._rs = _router_states.setdefault(
            (app_name, deployment_name or ""), _RouterState()
        )

    def __getstate__(self):
        d = dict(self.__dict__)
        d.pop("_rs", None)  # router state never crosses the wire
        return d

    def __setstate__(self, d):
        self.__dict__.update(d)
        self._rs = _router_states.setdefault(
            (self.app_name, self.deployment_name or ""), _RouterState()
        )

    @property
    def _replicas(self):
        return self._rs.replicas

    @property
    def _version(self):
        return self._rs.version

    @property
    def _counts(self):
        return self._rs.counts

    def _start_watch(self):
        """Push-based config updates (reference: LongPollClient in every
        router/proxy): one daemon thread per deployment long-polls the
        controller and swaps the shared replica set when the version
        moves."""
        rs = self._rs
        if rs.watch is not None:
            return
        import threading
        import time as _t

        app, dep = self.app_name, self.deployment_name

        def watch():
            ray = _ray()
            failures = 0
            while failures < 30:
                try:
                    ctrl = ray.get_actor(SERVE_CONTROLLER_NAME,
                                         namespace=SERVE_NAMESPACE)
                    info = ray.get(
                        ctrl.listen_for_change.remote(app, dep, rs.version),
                        timeout=45,
                    )
                    failures = 0
                except Exception:
                    failures += 1
                    _t.sleep(1.0)
                    continue
                if info is not None:
                    rs.replicas = info["replicas"]
                    rs.version = info["version"]
                    rs.counts = {i: 0 for i in range(len(info["replicas"]))}

        t = threading.Thread(target=watch, daemon=True,
                             name="serve-longpoll")
        t.start()
        rs.watch = t

    def _refresh(self):
        ray = _ray()
        ctrl = ray.get_actor(SERVE_CONTROLLER_NAME, namespace=SERVE_NAMESPACE)
        info = ray.get(
            ctrl.get_routing.remote(self.app_name, self.deployment_name)
        )
        if info is None:
            raise RuntimeError(
                f"no deployment {self.deployment_name} in app {self.app_name}"
            )
        rs = self._rs
        rs.replicas = info["replicas"]
        rs.version = info["version"]
        rs.counts = {i: 0 for i in range(len(rs.replicas))}
        if self.deployment_name != info["deployment"]:
            self.deployment_name = info["deployment"]
            # alias the resolved name so clones share this state
            _router_states[(self.app_name, self.deployment_name)] = rs

    def _pick(self) -> int:
        n = len(self._replicas)
        if n == 1:
            return 0
        i, j = random.sample(range(n), 2)
        return i if self._counts.get(i, 0) <= self._counts.get(j, 0) else j

    def remote(self, *args, **kwargs) -> DeploymentResponse:
        if not self._replicas:
            self._refresh()
        self._start_watch()
        for attempt in range(3):
            idx = self._pick()
            replica = self._replicas[idx]
            self._counts[idx] = self._counts.get(idx, 0) + 1
            try:
                if self._stream:
                    gen = replica.handle_request.options(
                        num_returns="streaming"
                    ).remote(self.method_name, args, kwargs)
                    self._decr_later(idx)
                    return DeploymentResponseGenerator(gen)
                ref = replica.handle_request.remote(
                    self.method_name, args, kwargs
                )
                resp = DeploymentResponse(ref)
                self._decr_later(idx)
                return resp
            except Exception:
                self._refresh()
        raise RuntimeError("could not route request")

    def _decr_later(self, idx):
        # decremented optimistically; precise per-request accounting
        # happens replica-side (get_stats)
        self._counts[idx] = max(0, self._counts.get(idx, 1) - 1)

    def options(self, *, method_name: Optional[str] = None,
                stream: Optional[bool] = None, **kwargs):
        return DeploymentHandle(
            self.app_name, self.deployment_name,
            method_name or self.method_name,
            self._stream if stream is None else stream,
        )

    def __getattr__(self, item):
        if item.startswith("_"):
            raise AttributeError(item)
        return self.options(method_name=item)

    def __reduce__(self):
        return (DeploymentHandle,
                (self.app_name, self.deployment_name, self.method_name))


# --------------------------------------------------------------------------
# HTTP proxy
# --------------------------------------------------------------------------


class ProxyActor:
    def __init__(self, port: int):
        self.port = port
        self._handles: Dict[str, DeploymentHandle] = {}
        self._streaming: Dict[str, bool] = {}
        self._server_task = None

    async def start_server(self):
        import uvicorn

        ray = _ray()

        async def asgi(scope, receive, send):
            if scope["type"] != "http":
                return
            body = b""
            while True:
                msg = await receive()
                body += msg.get("body", b"")
                if not msg.get("more_body"):
                    break
            path = scope["path"]
            loop = asyncio.get_running_loop()
            # sync ray calls must leave the event loop (executor thread)
            route = await loop.run_in_executor(None, self._resolve_route, path)
            if route is None:
                await _send_response(send, 404, {}, b'{"error":"not found"}')
                return
            prefix, app_name = route
            h = self._handles.get(app_name)
            if h is None:
                h = self._handles[app_name] = DeploymentHandle(app_name)
                await loop.run_in_executor(None, h._refresh)
            sub_path = path[len(prefix.rstrip("/")):] or "/"
            headers = {
                k.decode(): v.decode() for k, v in scope.get("headers", [])
            }
            if not h._replicas:
                await loop.run_in_executor(None, h._refresh)
            from ray_amd._core import runtime as _rt

            rt = _rt.global_runtime()
            qs = scope.get("query_string", b"").decode()
            last_exc = None
            for _attempt in range(2):
                idx = h._pick()
                replica = h._replicas[idx]
                try:
                    st_flag = self._streaming.get(app_name)
                    if st_flag is None:
                        fref = replica.is_http_streaming.remote()
                        st_flag = (await rt.get_async([fref], 30))[0]
                        self._streaming[app_name] = st_flag
                    if st_flag:
                        await self._proxy_stream(
                            send, replica, scope["method"], sub_path, qs,
                            headers, body, rt, loop,
                        )
                        return
                    ref = replica.handle_http.remote(
                        scope["method"], sub_path, qs, headers, body
                    )
                    status, hdrs, content = (await rt.get_async([ref], 120))[0]
                    await _send_response(send, status, hdrs, content)
                    return
                except Exception as e:  # replica died: refresh + retry
                    last_exc = e
                    try:
                        await loop.run_in_executor(None, h._refresh)
                    except Exception:
                        break
            await _send_response(
                send, 500, {}, f'{{"error":"{last_exc}"}}'.encode()
            )

        # Try the configured port; on EADDRINUSE fall back to an
        # ephemeral port (a second node's proxy in the single-machine
        # multi-node sim — on a real cluster every node binds the same
        # port on its own host). Returns the BOUND port (truthy).
        async def _guarded_serve(server):
            # uvicorn sys.exit()s on bind failure — a SystemExit escaping
            # a task would tear down the actor's event loop
            try:
                await server.serve()
            except (SystemExit, OSError):
                pass

        for try_port in (self.port, 0):
            config = uvicorn.Config(
                asgi, host="127.0.0.1", port=try_port, log_level="warning",
                loop="asyncio",
            )
            self._server = uvicorn.Server(config)
            self._server_task = asyncio.ensure_future(
                _guarded_serve(self._server))
            for _ in range(100):
                if self._server.started:
                    try:
                        self.port = (self._server.servers[0].sockets[0]
                                     .getsockname()[1])
                    except Exception:
                        pass
                    return self.port
                if self._server_task.done():
                    break  # bind failed — retry ephemeral
                await asyncio.sleep(0.05)
        return False

    def bound_port(self):
        return self.port

    async def _proxy_stream(self, send, replica, method, path, qs, headers,
                            body, rt, loop):
        """Forward a streaming replica response as chunked HTTP body
        parts — each yielded item flushes to the client immediately."""
        gen = replica.handle_http_stream.options(
            num_returns="streaming"
        ).remote(method, path, qs, headers, body)
        started = False
        try:
            while True:
                ref = await loop.run_in_executor(None, next, gen, None)
                if ref is None:
                    break
                item = (await rt.get_async([ref], 120))[0]
                if not started:
                    status, hdrs = item
                    await send({
                        "type": "http.response.start",
                        "status": status,
                        "headers": [
                            (k.encode(), v.encode())
                            for k, v in (hdrs or {}).items()
                            if k.lower() not in
                            ("content-length", "transfer-encoding")
                        ],
                    })
                    started = True
                else:
                    await send({
                        "type": "http.response.body", "body": item,
                        "more_body": True,
                    })
        except Exception as e:
            if not started:
                await _send_response(
                    send, 500, {}, f'{{"error":"{e}"}}'.encode()
                )
                return
        await send({"type": "http.response.body", "body": b"",
                    "more_body": False})

    def _resolve_route(self, path):
        # per-path TTL cache: a controller RPC per request would cap
        # HTTP throughput (reference: proxies watch route tables via
        # long-poll instead of resolving per request)
        cache = getattr(self, "_route_cache", None)
        if cache is None:
            cache = self._route_cache = {}
        hit = cache.get(path)
        now = time.time()
        if hit is not None and now - hit[1] < 2.0:
            return hit[0]
        ray = _ray()
        ctrl = ray.get_actor(SERVE_CONTROLLER_NAME, namespace=SERVE_NAMESPACE)
        route = ray.get(ctrl.resolve_route.remote(path))
        cache[path] = (route, now)
        if len(cache) > 4096:
            cache.clear()
        return route

    def ping(self):
        return "pong"


async def _send_response(send, status, headers, body):
    await send(
        {
            "type": "http.response.start",
            "status": status,
            "headers": [
                (k.encode(), v.encode())
                for k, v in (headers or {}).items()
                if k.lower() not in ("content-length", "transfer-encoding")
            ]
            + [(b"content-length", str(len(body)).encode())],
        }
    )
    await send({"type": "http.response.body", "body": body})


# --------------------------------------------------------------------------
# gRPC proxy (reference: _private/proxy.py:555 gRPCProxy)
# --------------------------------------------------------------------------


class GrpcRequest:
    """What a deployment method receives from the gRPC ingress."""

    def __init__(self, method: str, data: bytes):
        self.method = method
        self.data = data


class GrpcProxyActor:
    """gRPC ingress on a generic handler: any unary-unary call to
    /<app_name>/<Method> with bytes request/response (no compiled proto
    needed on either side — clients use bytes serializers). Routes the
    call to the named application's deployment handle; the deployment's
    method <Method> (or __call__) receives a GrpcRequest.

    Reference: serve gRPCProxy (proxy.py:555), re-designed without the
    user-servicer registration machinery: the generic handler covers
    arbitrary service/method names."""

    def __init__(self, port: int):
        self.port = port
        self._handles: Dict[str, DeploymentHandle] = {}
        self._server = None

    async def start_server(self):
        import grpc

        ray = _ray()
        outer = self

        class _Generic(grpc.GenericRpcHandler):
            def service(self, hcd):
                # method path "/<app>/<Method>"
                parts = hcd.method.strip("/").split("/")
                if len(parts) != 2:
                    return None
                app_name, method = parts

                async def unary_unary(request: bytes, context):
                    return await outer._route(app_name, method, request)

                return grpc.unary_unary_rpc_method_handler(
                    unary_unary,
                    request_deserializer=None,
                    response_serializer=None,
                )

        server = grpc.aio.server()
        server.add_generic_rpc_handlers((_Generic(),))
        self.port = server.add_insecure_port(f"127.0.0.1:{self.port}")
        await server.start()
        self._server = server
        return self.port

    async def _route(self, app_name: str, method: str, data: bytes) -> bytes:
        from ray_amd._core import runtime as _rtmod

        rt = _rtmod.global_runtime()
        loop = asyncio.get_running_loop()
        h = self._handles.get(app_name)
        if h is None:
            h = self._handles[app_name] = DeploymentHandle(app_name)
            await loop.run_in_executor(None, h._refresh)
        if not h._replicas:
            await loop.run_in_executor(None, h._refresh)
        last = None
        for _ in range(2):
            try:
                idx = h._pick()
                replica = h._replicas[idx]
                ref = replica.handle_grpc.remote(method, data)
                return (await rt.get_async([ref], 120))[0]
            except Exception as e:
                last = e
                try:
                    await loop.run_in_executor(None, h._refresh)
                except Exception:
                    break
        raise RuntimeError(f"grpc route to {app_name} failed: {last}")


# --------------------------------------------------------------------------
# module-level API
# --------------------------------------------------------------------------

_http_port = 8000
_grpc_port: Optional[int] = None
_started = False


def start(detached: bool = True, http_options: Optional[dict] = None,
          grpc_options: Optional[dict] = None, **kw):
    global _http_port, _grpc_port, _started
    if http_options:
        _http_port = http_options.get("port", _http_port)
    if grpc_options:
        _grpc_port = grpc_options.get("port", 9000)
    _ensure_controller()
    if _grpc_port is not None:
        _ensure_grpc_proxy(_grpc_port)
    _started = True


SERVE_GRPC_PROXY_NAME = "SERVE_GRPC_PROXY"


def _ensure_grpc_proxy(port: int):
    ray = _ray()
    try:
        return ray.get_actor(SERVE_GRPC_PROXY_NAME, namespace=SERVE_NAMESPACE)
    except ValueError:
        P = ray.remote(GrpcProxyActor)
        proxy = P.options(
            name=SERVE_GRPC_PROXY_NAME, namespace=SERVE_NAMESPACE,
            num_cpus=0.1, max_concurrency=64,
        ).remote(port)
        bound = ray.get(proxy.start_server.remote(), timeout=30)
        if not bound:
            raise RuntimeError("serve gRPC proxy failed to start")
        return proxy


def _ensure_controller():
    ray = _ray()
    try:
        return ray.get_actor(SERVE_CONTROLLER_NAME, namespace=SERVE_NAMESPACE)
    except ValueError:
        Ctrl = ray.remote(ServeController)
        ctrl = Ctrl.options(
            name=SERVE_CONTROLLER_NAME, namespace=SERVE_NAMESPACE,
            num_cpus=0.1, max_concurrency=16,
        ).remote()
        ray.get(ctrl.ping.remote())
        return ctrl


def _ensure_proxy(port: int):
    ray = _ray()
    try:
        return ray.get_actor(SERVE_PROXY_NAME, namespace=SERVE_NAMESPACE)
    except ValueError:
        P = ray.remote(ProxyActor)
        proxy = P.options(
            name=SERVE_PROXY_NAME, namespace=SERVE_NAMESPACE, num_cpus=0.1,
            max_concurrency=64,
        ).remote(port)
        ok = ray.get(proxy.start_server.remote(), timeout=30)
        if not ok:
            raise RuntimeError("serve HTTP proxy failed to start")
        return proxy


def _ensure_node_proxies(port: int) -> Dict[str, int]:
    """One HTTP proxy per alive node (reference: ProxyStateManager in
    _private/proxy_state.py starts a proxy actor on every node). The
    head-node singleton keeps the legacy SERVE_PROXY_ACTOR name; the
    others are SERVE_PROXY_ACTOR:<node_id> hard-pinned to their node.
    Returns {node_id_hex: bound_port}."""
    ray = _ray()
    from ray_amd.util.scheduling_strategies import (
        NodeAffinitySchedulingStrategy,
    )

    out: Dict[str, int] = {}
    for n in ray.nodes():
        if not n["Alive"]:
            continue
        nid = n["NodeID"]
        aname = f"{SERVE_PROXY_NAME}:{nid}"
        try:
            proxy = ray.get_actor(aname, namespace=SERVE_NAMESPACE)
            out[nid] = ray.get(proxy.bound_port.remote(), timeout=30)
            continue
        except ValueError:
            pass
        P = ray.remote(ProxyActor)
        proxy = P.options(
            name=aname, namespace=SERVE_NAMESPACE, num_cpus=0.1,
            max_concurrency=64,
            scheduling_strategy=NodeAffinitySchedulingStrategy(
                nid, soft=False),
        ).remote(port)
        bound = ray.get(proxy.start_server.remote(), timeout=30)
        if not bound:
            raise RuntimeError(f"serve proxy failed to start on node {nid}")
        out[nid] = bound
    return out


def proxy_ports() -> Dict[str, int]:
    """Bound HTTP proxy port per node id (reference: serve.status()
    proxies field)."""
    ray = _ray()
    out: Dict[str, int] = {}
    for n in ray.nodes():
        if not n["Alive"]:
            continue
        try:
            p = ray.get_actor(f"{SERVE_PROXY_NAME}:{n['NodeID']}",
                              namespace=SERVE_NAMESPACE)
            out[n["NodeID"]] = ray.get(p.bound_port.remote(), timeout=30)
        except Exception:
            pass
    return out


def run(app: Application, *, name: str = "default", route_prefix: str = "/",
        blocking: bool = False, _local_testing_mode: bool = False,
        http: bool = True, port: Optional[int] = None) -> DeploymentHandle:
    import cloudpickle

    ray = _ray()
    ctrl = _ensure_controller()
    nodes = app._collect()
    # assign unique per-node deployment names (reference suffixes
    # duplicate bindings: Adder, Adder_1, ...)
    node_names: Dict[int, str] = {}
    used: Dict[str, int] = {}
    for node in nodes:
        base = node.deployment.name
        k = used.get(base, 0)
        node_names[id(node)] = base if k == 0 else f"{base}_{k}"
        used[base] = k + 1
    specs = []
    for node in nodes:
        d = node.deployment

        def resolve(v):
            if isinstance(v, Application):
                return DeploymentHandle(name, node_names[id(v.root)])
            if isinstance(v, DeploymentNode):
                return DeploymentHandle(name, node_names[id(v)])
            return v

        args = tuple(resolve(a) for a in node.args)
        kwargs = {k: resolve(v) for k, v in node.kwargs.items()}
        asc = None
        if d.autoscaling_config:
            a = d.autoscaling_config
            asc = {
                "min_replicas": a.min_replicas,
                "max_replicas": a.max_replicas,
                "target_ongoing_requests": a.target_ongoing_requests,
                "downscale_delay_s": a.downscale_delay_s,
            }
        specs.append(
            {
                "name": node_names[id(node)],
                "num_replicas": d.num_replicas,
                "max_ongoing_requests": d.max_ongoing_requests,
                "ray_actor_options": d.ray_actor_options,
                "user_config": d.user_config,
                "autoscaling": asc,
                "pg_bundles": d.placement_group_bundles,
                "pg_strategy": d.placement_group_strategy,
                "payload": cloudpickle.dumps(
                    (d.func_or_class, args, kwargs)
                ),
            }
        )
    ingress_name = node_names[id(app.root)]
    ray.get(
        ctrl.deploy_application.remote(name, route_prefix, specs, ingress_name),
        timeout=120,
    )
    if http:
        _ensure_proxy(port or _http_port)
        try:
            alive = [n for n in ray.nodes() if n["Alive"]]
            if len(alive) > 1:
                _ensure_node_proxies(port or _http_port)
        except Exception:
            pass  # single-node path keeps the head proxy only
    h = DeploymentHandle(name)
    h._refresh()
    return h


def get_app_handle(name: str = "default") -> DeploymentHandle:
    h = DeploymentHandle(name)
    h._refresh()
    return h


def get_deployment_handle(deployment_name: str, app_name: str = "default"
                          ) -> DeploymentHandle:
    h = DeploymentHandle(app_name, deployment_name)
    h._refresh()
    return h


def status():
    ray = _ray()
    ctrl = _ensure_controller()
    out = ray.get(ctrl.status.remote())
    # reference serve.status(): proxies section (node -> state/port)
    try:
        out["proxies"] = {
            nid: {"status": "HEALTHY", "port": port}
            for nid, port in proxy_ports().items()
        }
    except Exception:
        pass
    return out


def delete(name: str, _blocking: bool = True):
    ray = _ray()
    ctrl = _ensure_controller()
    ray.get(ctrl.delete_application.remote(name))


def shutdown():
    ray = _ray()
    names = [SERVE_PROXY_NAME, SERVE_CONTROLLER_NAME]
    try:
        names += [f"{SERVE_PROXY_NAME}:{n['NodeID']}" for n in ray.nodes()]
    except Exception:
        pass
    for n in names:
        try:
            a = ray.get_actor(n, namespace=SERVE_NAMESPACE)
            ray.kill(a)
        except Exception:
            pass
    _router_states.clear()  # drop stale routing (+ dead watch threads)


# --------------------------------------------------------------------------
# serve.multiplexed (reference: serve/multiplex.py — per-replica LRU of
# loaded models keyed by model id)
# --------------------------------------------------------------------------

_mux_ctx_model_id: Optional[str] = None


def get_multiplexed_model_id() -> str:
    return _mux_ctx_model_id or ""


def multiplexed(_func=None, *, max_num_models_per_replica: int = 3):
    """Decorator for an async model-loader method; calls are routed with
    handle.options(multiplexed_model_id=...) and the replica keeps an
    LRU of loaded models."""

    def dec(loader):
        import collections

        cache: "collections.OrderedDict" = collections.OrderedDict()

        async def wrapper(self_arg, model_id: str):
            global _mux_ctx_model_id
            if model_id in cache:
                cache.move_to_end(model_id)
                return cache[model_id]
            _mux_ctx_model_id = model_id
            try:
                model = loader(self_arg, model_id)
                if asyncio.iscoroutine(model):
                    model = await model
            finally:
                _mux_ctx_model_id = None
            cache[model_id] = model
            while len(cache) > max_num_models_per_replica:
                evicted_id, evicted = cache.popitem(last=False)
                dtor = getattr(evicted, "__del__", None)
                del evicted
            return model

        wrapper._serve_multiplexed = True
        return wrapper

    if _func is not None:
        return dec(_func)
    return dec


# --------------------------------------------------------------------------
# serve.batch
# --------------------------------------------------------------------------


def batch(_func=None, *, max_batch_size: int = 10,
          batch_wait_timeout_s: float = 0.01):
    """Dynamic request batching (reference: serve/batching.py)."""

    def dec(func):
        state = {"queue": None, "task": None}

        async def flusher(queue):
            while True:
                items = [await queue.get()]
                deadline = asyncio.get_running_loop().time() + batch_wait_timeout_s
                while len(items) < max_batch_size:
                    to = deadline - asyncio.get_running_loop().time()
                    if to <= 0:
                        break
                    try:
                        items.append(
                            await asyncio.wait_for(queue.get(), timeout=to)
                        )
                    except asyncio.TimeoutError:
                        break
                args = [it[0] for it in items]
                futs = [it[1] for it in items]
                try:
                    self_arg = items[0][2]
                    if self_arg is not None:
                        results = await func(self_arg, args)
                    else:
                        results = await func(args)
                    for f, r in zip(futs, results):
                        if not f.done():
                            f.set_result(r)
                except Exception as e:
                    for f in futs:
                        if not f.done():
                            f.set_exception(e)

        async def wrapper(*call_args):
            if state["queue"] is None:
                state["queue"] = asyncio.Queue()
                state["task"] = asyncio.ensure_future(flusher(state["queue"]))
            if len(call_args) == 2:
                self_arg, item = call_args
            else:
                self_arg, item = None, call_args[0]
            fut = asyncio.get_running_loop().create_future()
            await state["queue"].put((item, fut, self_arg))
            return await fut

        wrapper._is_serve_batch = True
        return wrapper

    if _func is not None:
        return dec(_func)
    return dec
