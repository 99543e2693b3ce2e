"""Public API: init/shutdown, @remote, get/put/wait, actors.

API-compatible with the reference's python surface
(python/ray/_private/worker.py:1439 ray.init, :3730 ray.remote,
:2856/:3025/:3081 get/put/wait; python/ray/actor.py ActorClass/
ActorHandle) — implemented on ray_amd's own runtime.
"""
from __future__ import annotations

import functools
import hashlib
import os
import threading
from typing import Any, Dict, List, Optional, Sequence, Union

import cloudpickle

from . import exceptions as exc
from ._core import node as _node
from ._core import runtime as _rt
from ._core.runtime import ObjectRef

_init_lock = threading.Lock()
_cluster: Optional[_node.LocalCluster] = None
_namespace = "default"


def _atexit_shutdown():
    try:
        shutdown(_exiting_interpreter=True)
    except Exception:
        pass


import atexit  # noqa: E402

atexit.register(_atexit_shutdown)

DEFAULT_TASK_OPTIONS = dict(num_cpus=1, num_gpus=0, num_returns=1, max_retries=3)
# Reference semantics (actor.py): an actor with no explicit resource
# request reserves NOTHING while alive (scheduled by 1-CPU availability
# but released after start); explicit num_cpus/num_gpus are reserved
# for the actor's lifetime.
DEFAULT_ACTOR_OPTIONS = dict(
    num_cpus=None, num_gpus=0, max_restarts=0, max_concurrency=1, lifetime=None
)


class RayContext:
    def __init__(self, session_dir, node_id):
        self.session_dir = session_dir
        self.node_id = node_id
        self.address_info = {"session_dir": session_dir, "node_id": node_id.hex()}
        self.dashboard_url = None

    def __enter__(self):
        return self

    def __exit__(self, *a):
        shutdown()

    def disconnect(self):
        shutdown()


def init(
    address: Optional[str] = None,
    *,
    num_cpus: Optional[float] = None,
    num_gpus: Optional[float] = None,
    resources: Optional[Dict[str, float]] = None,
    object_store_memory: Optional[int] = None,
    namespace: Optional[str] = None,
    ignore_reinit_error: bool = False,
    runtime_env: Optional[dict] = None,
    labels: Optional[dict] = None,
    log_to_driver: bool = True,
    configure_logging: bool = True,
    include_dashboard: Optional[bool] = None,
    dashboard_host: str = "127.0.0.1",
    dashboard_port: Optional[int] = None,
    _system_config: Optional[dict] = None,
    **kwargs,
) -> RayContext:
    global _cluster, _namespace
    if address is None:
        # job drivers / tooling inherit the cluster address from the
        # environment (reference: RAY_ADDRESS)
        address = os.environ.get("RAY_AMD_ADDRESS") or None
    with _init_lock:
        if _rt.is_initialized():
            if ignore_reinit_error:
                rt = _rt.global_runtime()
                return RayContext(rt.session_dir, rt.node_id)
            raise RuntimeError(
                "ray_amd.init() called twice; pass ignore_reinit_error=True"
            )
        if namespace:
            _namespace = namespace
        if runtime_env and runtime_env.get("env_vars"):
            os.environ.update(
                {str(k): str(v) for k, v in runtime_env["env_vars"].items()}
            )
        if address and (
            address.startswith("ray_amd://") or address.startswith("ray://")
        ):
            # Ray-Client mode: proxy every op to a ClientServer on the
            # cluster (reference: ray.init("ray://...") util/client/)
            from .client import ClientRuntime

            crt = ClientRuntime(address.split("://", 1)[1],
                                namespace=_namespace)
            crt.connect()
            _rt.set_global_runtime(crt)
            return RayContext(crt.session_dir, crt.node_id)
        if address in (None, "local"):
            cluster = _node.start_local_cluster(
                num_cpus=num_cpus,
                num_gpus=num_gpus,
                resources=resources,
                object_store_memory=object_store_memory,
                labels=labels,
            )
            _cluster = cluster
            info = {
                "gcs_addr": cluster.gcs_addr,
                "raylet_addr": cluster.raylet_addr,
                "node_id": cluster.node_id.hex(),
                "session_dir": cluster.session_dir,
            }
        else:
            info = _node.find_session(address)
            os.environ["RAY_AMD_SHM_DIR"] = _node.session_shm_dir(
                info["session_dir"]
            )
        rt = _rt.CoreRuntime(
            "driver",
            info["session_dir"],
            info["gcs_addr"],
            info["raylet_addr"],
            bytes.fromhex(info["node_id"]),
        )
        rt.start_driver()
        _rt.set_global_runtime(rt)
        rt.job_id = rt.gcs_call("next_job_id", {})
        return RayContext(info["session_dir"], bytes.fromhex(info["node_id"]))


def shutdown(_exiting_interpreter: bool = False):
    global _cluster
    with _init_lock:
        try:
            from .util.usage_stats import usage_stats_enabled, write_report

            if _rt.is_initialized() and usage_stats_enabled():
                write_report(_rt.global_runtime().session_dir)
        except Exception:
            pass
        try:
            if _rt.is_initialized():
                rt = _rt.global_runtime()
                rt.shutdown()
        except Exception:
            pass
        # only the driver that STARTED the cluster tears it down;
        # connected drivers just disconnect
        if _cluster is not None:
            try:
                _cluster.shutdown()
            except Exception:
                pass
            _cluster = None


def is_initialized() -> bool:
    return _rt.is_initialized()


# --------------------------------------------------------------------------
# tasks
# --------------------------------------------------------------------------


class RemoteFunction:
    def __init__(self, fn, options: dict):
        self._function = fn
        self._options = {**DEFAULT_TASK_OPTIONS, **options}
        self._pickled = None
        self._fn_id = None
        self._exported_rt = None
        functools.update_wrapper(self, fn)

    def _ensure_exported(self, rt):
        if self._pickled is None:
            self._pickled = cloudpickle.dumps(self._function)
            self._fn_id = hashlib.sha1(self._pickled).digest()
        if self._exported_rt is not rt:
            rt._call_sync(rt._export_function(self._fn_id, self._pickled))
            self._exported_rt = rt

    def remote(self, *args, **kwargs):
        return self._remote(args, kwargs, self._options)

    def options(self, **opts):
        merged = {**self._options, **opts}
        parent = self

        class _Opted:
            def remote(self, *args, **kwargs):
                return parent._remote(args, kwargs, merged)

        return _Opted()

    def _remote(self, args, kwargs, opts):
        rt = _rt.global_runtime()
        self._ensure_exported(rt)
        opts = _normalize_scheduling(opts)
        if opts.get("runtime_env"):
            opts = dict(opts)
            opts["runtime_env"] = _stage_runtime_env(opts["runtime_env"])
        name = opts.get("name") or getattr(self._function, "__name__", "fn")
        refs = rt.submit_task(self._pickled, self._fn_id, name, (args, kwargs), opts)
        if opts.get("num_returns", 1) == 1:
            return refs[0]
        return refs

    def __call__(self, *a, **k):
        raise TypeError(
            "Remote functions cannot be called directly; use .remote()"
        )

    def __reduce__(self):
        # never pickle the runtime/export caches (they hold sockets)
        return (_rebuild_remote_function,
                (cloudpickle.dumps(self._function), self._options))

    def bind(self, *args, **kwargs):
        from .dag import FunctionNode

        return FunctionNode(self, args, kwargs)


def _rebuild_remote_function(fn_bytes, options):
    return RemoteFunction(cloudpickle.loads(fn_bytes), options)


def _rebuild_actor_class(cls_bytes, options):
    return ActorClass(cloudpickle.loads(cls_bytes), options)


def _stage_runtime_env(runtime_env: Optional[dict]) -> Optional[dict]:
    """Copy working_dir / py_modules into the session dir so every
    worker on the node can import them (reference: runtime_env
    working_dir upload; conda/pip are N/A offline)."""
    if not runtime_env:
        return runtime_env
    needs = runtime_env.get("working_dir") or runtime_env.get("py_modules")
    if not needs:
        return runtime_env
    import hashlib as _h
    import shutil as _sh

    rt = _rt.global_runtime()
    if getattr(rt, "is_client", False):
        raise NotImplementedError(
            "runtime_env working_dir/py_modules over Ray-Client mode "
            "needs a file upload channel (next round); env_vars work"
        )
    out = dict(runtime_env)
    base = os.path.join(rt.session_dir, "runtime_env")
    os.makedirs(base, exist_ok=True)

    def stage(path):
        path = os.path.abspath(path)
        tag = _h.sha1(path.encode()).hexdigest()[:12]
        dest = os.path.join(base, tag)
        if not os.path.exists(dest):
            _sh.copytree(path, dest)
        return dest

    if out.get("working_dir"):
        out["working_dir"] = stage(out["working_dir"])
    if out.get("py_modules"):
        out["py_modules"] = [stage(p) for p in out["py_modules"]]
    return out


def _normalize_scheduling(opts: dict) -> dict:
    opts = dict(opts)
    acc = opts.pop("accelerator_type", None)
    if acc:
        # reference: accelerator_type=X adds a tiny demand on the
        # node's accelerator_type:X resource (util/accelerators)
        res = dict(opts.get("resources") or {})
        res[f"accelerator_type:{acc}"] = 0.001
        opts["resources"] = res
    strat = opts.get("scheduling_strategy")
    if strat is not None and hasattr(strat, "placement_group"):
        pg = strat.placement_group
        opts["placement_group"] = (
            pg.id,
            getattr(strat, "placement_group_bundle_index", None),
        )
    elif strat is not None and hasattr(strat, "node_id"):
        # NodeAffinitySchedulingStrategy (reference:
        # policy/node_affinity_scheduling_policy.cc)
        opts["node_affinity"] = (str(strat.node_id), bool(strat.soft))
    elif strat is not None and hasattr(strat, "hard"):
        # NodeLabelSchedulingStrategy
        opts["label_selector"] = {
            "hard": dict(strat.hard or {}),
            "soft": dict(strat.soft or {}),
        }
    elif opts.get("placement_group") is not None and not isinstance(
        opts.get("placement_group"), tuple
    ):
        pg = opts["placement_group"]
        opts["placement_group"] = (
            pg.id,
            opts.get("placement_group_bundle_index"),
        )
    return opts


# --------------------------------------------------------------------------
# actors
# --------------------------------------------------------------------------


class ActorMethod:
    def __init__(self, handle: "ActorHandle", name: str, defaults=None):
        self._handle = handle
        self._name = name
        # per-method defaults from @ray.method(...) on the class
        self._defaults = dict(defaults or {})

    def remote(self, *args, **kwargs):
        return self._remote(args, kwargs, {})

    def options(self, **opts):
        parent = self

        class _Opted:
            def remote(self, *args, **kwargs):
                return parent._remote(args, kwargs, opts)

        return _Opted()

    def _remote(self, args, kwargs, opts):
        rt = _rt.global_runtime()
        merged = {**self._defaults, **opts}
        merged.setdefault("num_returns", 1)
        merged.setdefault("max_task_retries", self._handle._max_task_retries)
        num_returns = merged["num_returns"]
        if self._handle._tensor_transport:
            # sender-side offload (reference: rdt_manager __ray_send__):
            # GPU tensors in args ship as GPU-store refs
            from .experimental import rdt as _rdt

            if _rdt.has_cuda_tensors((args, kwargs)):
                args, _ = _rdt.offload_tensors(list(args))
                args = tuple(args)
                kwargs, _ = _rdt.offload_tensors(kwargs)
        refs = rt.submit_actor_task(
            self._handle._actor_id,
            self._name,
            (args, kwargs),
            merged,
        )
        if num_returns == "streaming":
            return refs
        if num_returns == 1:
            return refs[0]
        return refs


class ActorHandle:
    def __init__(self, actor_id: bytes, class_name: str = "Actor",
                 method_options=None, tensor_transport=None,
                 max_task_retries=0):
        self._actor_id = actor_id
        self._class_name = class_name
        self._method_options = method_options or {}
        self._tensor_transport = tensor_transport
        self._max_task_retries = max_task_retries

    def __getattr__(self, item):
        if item.startswith("_") and item not in (
            "__ray_apply__", "__ray_terminate__"
        ):
            raise AttributeError(item)
        return ActorMethod(self, item, self._method_options.get(item))

    def __repr__(self):
        return f"Actor({self._class_name}, {self._actor_id.hex()})"

    def __reduce__(self):
        return (_deserialize_handle,
                (self._actor_id, self._class_name, self._method_options,
                 self._tensor_transport, self._max_task_retries))

    def __hash__(self):
        return hash(self._actor_id)

    def __eq__(self, other):
        return (
            isinstance(other, ActorHandle) and other._actor_id == self._actor_id
        )

    def _actor_ref(self):
        return self._actor_id


def _deserialize_handle(actor_id, class_name, method_options=None,
                        tensor_transport=None, max_task_retries=0):
    return ActorHandle(actor_id, class_name, method_options,
                       tensor_transport, max_task_retries)


class ActorClass:
    def __init__(self, cls, options: dict):
        self._cls = cls
        self._options = {**DEFAULT_ACTOR_OPTIONS, **options}
        self._pickled = None
        self._key = None
        self._pickled_conc = None

    def _ensure_pickled(self, max_concurrency):
        if self._pickled is None or self._pickled_conc != max_concurrency:
            self._pickled = cloudpickle.dumps((self._cls, max_concurrency))
            self._key = hashlib.sha1(self._pickled).digest()
            self._pickled_conc = max_concurrency
        return self._pickled, self._key

    def remote(self, *args, **kwargs):
        return self._remote(args, kwargs, self._options)

    def options(self, **opts):
        merged = {**self._options, **opts}
        parent = self

        class _Opted:
            def remote(self, *args, **kwargs):
                return parent._remote(args, kwargs, merged)

        return _Opted()

    def _remote(self, args, kwargs, opts):
        rt = _rt.global_runtime()
        opts = _normalize_scheduling(opts)
        opts = dict(opts)
        if opts.get("runtime_env"):
            opts["runtime_env"] = _stage_runtime_env(opts["runtime_env"])
        opts.setdefault("namespace", _namespace)
        opts["class_name"] = self._cls.__name__
        mc = opts.get("max_concurrency", 1)
        pickled, key = self._ensure_pickled(mc)
        actor_id = rt.create_actor(key, pickled, opts, (args, kwargs))
        return ActorHandle(actor_id, self._cls.__name__,
                           self._method_options(),
                           opts.get("tensor_transport"),
                           opts.get("max_task_retries", 0))

    def _method_options(self) -> dict:
        """Collect @ray.method(...) per-method option dicts off the class
        (reference: actor.py __ray_method_options__ on ActorMethod)."""
        out = {}
        for name in dir(self._cls):
            if name.startswith("__"):
                continue
            try:
                attr = getattr(self._cls, name)
            except Exception:
                continue
            mo = getattr(attr, "__ray_method_options__", None)
            if mo:
                out[name] = dict(mo)
        return out

    def bind(self, *args, **kwargs):
        from .dag import ClassNode

        return ClassNode(self, args, kwargs)

    def __reduce__(self):
        return (_rebuild_actor_class,
                (cloudpickle.dumps(self._cls), self._options))

    def __call__(self, *a, **k):
        raise TypeError("Actors cannot be instantiated directly; use .remote()")


def remote(*args, **kwargs):
    """@ray.remote decorator for functions and classes."""

    def make(obj):
        if isinstance(obj, type):
            return ActorClass(obj, kwargs)
        return RemoteFunction(obj, kwargs)

    if len(args) == 1 and not kwargs and (callable(args[0]) or isinstance(args[0], type)):
        return make(args[0])
    if args:
        raise TypeError("@remote takes keyword arguments only")
    return make


def method(**kwargs):
    """@ray.method decorator (num_returns on actor methods)."""

    def dec(f):
        f.__ray_method_options__ = kwargs
        return f

    return dec


# --------------------------------------------------------------------------
# get / put / wait / kill / cancel
# --------------------------------------------------------------------------


def get(refs: Union[ObjectRef, Sequence[ObjectRef]], *, timeout: Optional[float] = None):
    from .dag import DAGFuture

    if isinstance(refs, DAGFuture):  # compiled-DAG result handle
        return refs.get(timeout if timeout is not None else 60.0)
    rt = _rt.global_runtime()
    if isinstance(refs, ObjectRef):
        return rt.get_sync([refs], timeout)[0]
    if isinstance(refs, list):
        if not refs:
            return []
        if not all(isinstance(r, ObjectRef) for r in refs):
            raise TypeError("ray.get() expects ObjectRef or list of ObjectRefs")
        return rt.get_sync(list(refs), timeout)
    raise TypeError(f"ray.get() got {type(refs)}")


def put(value: Any, *, _owner=None) -> ObjectRef:
    if isinstance(value, ObjectRef):
        raise TypeError("Calling ray.put on an ObjectRef is not allowed")
    return _rt.global_runtime().put(value, _owner)


def wait(
    refs: List[ObjectRef],
    *,
    num_returns: int = 1,
    timeout: Optional[float] = None,
    fetch_local: bool = True,
):
    if isinstance(refs, ObjectRef):
        raise TypeError("ray.wait() expects a list of ObjectRefs")
    if len(set(refs)) != len(refs):
        raise ValueError("ray.wait() got duplicate ObjectRefs")
    if num_returns > len(refs):
        raise ValueError("num_returns > number of refs")
    if num_returns <= 0:
        raise ValueError("num_returns <= 0")
    return _rt.global_runtime().wait_sync(refs, num_returns, timeout, fetch_local)


def kill(actor: ActorHandle, *, no_restart: bool = True):
    if not isinstance(actor, ActorHandle):
        raise TypeError("ray.kill() expects an ActorHandle")
    _rt.global_runtime().kill_actor(actor._actor_id, no_restart)


def cancel(ref: ObjectRef, *, force: bool = False, recursive: bool = True):
    """Best-effort cancellation: tasks not yet started are dropped and
    their refs resolve to TaskCancelledError (reference semantics for
    non-force cancel; running tasks are not interrupted)."""
    _rt.global_runtime().cancel_task(ref.id)


def get_actor(name: str, namespace: Optional[str] = None) -> ActorHandle:
    rt = _rt.global_runtime()
    r = rt.gcs_call(
        "resolve_actor",
        {"name": name, "namespace": namespace or _namespace, "wait": False},
    )
    if r.get("state") in ("NOT_FOUND", "DEAD") or r.get("actor_id") is None:
        raise ValueError(f"Failed to look up actor with name '{name}'")
    return ActorHandle(bytes(r["actor_id"]))


# --------------------------------------------------------------------------
# cluster info
# --------------------------------------------------------------------------


def nodes() -> List[dict]:
    rt = _rt.global_runtime()
    table = rt.gcs_call("node_table", {})
    out = []
    for n in table:
        out.append(
            {
                "NodeID": bytes(n["node_id"]).hex(),
                "Alive": n["alive"],
                "Draining": n.get("draining", False),
                "Resources": n["resources_total"],
                "Available": n["resources_available"],
                "Labels": n.get("labels", {}),
                "Address": n["addr"],
            }
        )
    return out


def drain_node(node_id: str, graceful: bool = True,
               deadline_s: float = 30.0) -> bool:
    """Drain a node (reference: DrainNode RPC / autoscaler scale-down).
    Graceful: the scheduler stops placing work on it immediately, and
    the node is declared dead once its leases finish or the deadline
    passes. graceful=False kills it from the cluster view now."""
    rt = _rt.global_runtime()
    return bool(rt.gcs_call("drain_node", {
        "node_id": bytes.fromhex(node_id), "graceful": graceful,
        "deadline_s": deadline_s,
    }))


def cluster_resources() -> Dict[str, float]:
    total: Dict[str, float] = {}
    for n in nodes():
        if not n["Alive"]:
            continue
        for k, v in n["Resources"].items():
            total[k] = total.get(k, 0) + v
    return total


def available_resources() -> Dict[str, float]:
    total: Dict[str, float] = {}
    for n in nodes():
        if not n["Alive"]:
            continue
        for k, v in n["Available"].items():
            total[k] = total.get(k, 0) + v
    return total


def timeline(filename: Optional[str] = None) -> Optional[list]:
    """Chrome-trace task events (reference: ray.timeline); returns the
    event list, and writes JSON when filename is given."""
    import json as _json

    from .util import state as _state

    events = []
    for t in _state.list_tasks():
        events.append({
            "cat": "task", "name": t["name"], "ph": "X",
            "ts": t["start_time_ms"] * 1000,
            "dur": (t["end_time_ms"] - t["start_time_ms"]) * 1000,
            "pid": t["worker_pid"], "tid": t["worker_pid"],
        })
    if filename:
        with open(filename, "w") as f:
            _json.dump(events, f)
        return None
    return events


def get_gpu_ids() -> List[int]:
    ids_env = os.environ.get("RAY_AMD_GPU_IDS")
    if ids_env:
        return [int(x) for x in ids_env.split(",") if x]
    try:
        from ._core.worker import current_task_context

        return list(current_task_context().gpu_ids)
    except Exception:
        return []


class RuntimeContext:
    @property
    def namespace(self):
        return _namespace

    def get_node_id(self) -> str:
        return _rt.global_runtime().node_id.hex()

    def get_actor_id(self) -> Optional[str]:
        try:
            from ._core.worker import current_task_context

            aid = current_task_context().actor_id
            return aid.hex() if aid else None
        except Exception:
            return None

    def get_task_id(self) -> Optional[str]:
        try:
            from ._core.worker import current_task_context

            tid = current_task_context().task_id
            return tid.hex() if tid else None
        except Exception:
            return None

    def get_job_id(self) -> str:
        return str(_rt.global_runtime().job_id)

    def get_worker_id(self) -> str:
        return _rt.global_runtime().worker_id.hex()

    def get_assigned_resources(self):
        return {}

    @property
    def gcs_address(self):
        return _rt.global_runtime().gcs_addr


def get_runtime_context() -> RuntimeContext:
    return RuntimeContext()
