"""DAG API (reference: python/ray/dag/ — DAGNode.bind, InputNode,
MultiOutputNode, experimental_compile for accelerated DAGs).

Full bind/execute composition over tasks and actor methods, and a
channel-compiled execution mode: experimental_compile() launches one
persistent loop per participating actor (via __ray_apply__) that reads
shm channels (experimental/channel.py), executes the bound methods in
topo order, and writes result channels — steady-state executions do
zero task submissions (reference: dag/compiled_dag_node.py). DAGs with
plain function nodes fall back to per-call submission.
"""
from __future__ import annotations

from typing import Any, Dict, List, Optional


class DAGNode:
    def execute(self, *args, **kwargs):
        raise NotImplementedError

    def experimental_compile(self, **kwargs) -> "CompiledDAG":
        return CompiledDAG(self)


class InputNode(DAGNode):
    """Placeholder for the DAG input (reference: dag/input_node.py).
    Usable as a context manager: `with InputNode() as inp:`"""

    def __enter__(self):
        return self

    def __exit__(self, *a):
        pass

    def execute(self, *args, **kwargs):
        raise RuntimeError("InputNode is resolved at DAG execution time")


class FunctionNode(DAGNode):
    def __init__(self, fn, args, kwargs):
        self._fn = fn
        self._args = args
        self._kwargs = kwargs

    def execute(self, *input_args, _cache: Dict = None, **input_kwargs):
        cache = _cache if _cache is not None else {}
        args = [
            _resolve(a, input_args, input_kwargs, cache) for a in self._args
        ]
        kwargs = {
            k: _resolve(v, input_args, input_kwargs, cache)
            for k, v in self._kwargs.items()
        }
        return self._fn.remote(*args, **kwargs)


class ClassMethodNode(DAGNode):
    def __init__(self, actor_node, method_name, args, kwargs):
        self._actor_node = actor_node
        self._method = method_name
        self._args = args
        self._kwargs = kwargs

    def execute(self, *input_args, _cache: Dict = None, **input_kwargs):
        cache = _cache if _cache is not None else {}
        actor = self._actor_node._get_actor()
        args = [
            _resolve(a, input_args, input_kwargs, cache) for a in self._args
        ]
        kwargs = {
            k: _resolve(v, input_args, input_kwargs, cache)
            for k, v in self._kwargs.items()
        }
        return getattr(actor, self._method).remote(*args, **kwargs)


class ClassNode(DAGNode):
    def __init__(self, cls, args, kwargs):
        self._cls = cls
        self._args = args
        self._kwargs = kwargs
        self._actor = None

    def _get_actor(self):
        if self._actor is None:
            self._actor = self._cls.remote(*self._args, **self._kwargs)
        return self._actor

    def execute(self, *a, **k):
        return self._get_actor()

    def __getattr__(self, item):
        if item.startswith("_"):
            raise AttributeError(item)
        node = self

        class _MethodBinder:
            def bind(self, *args, **kwargs):
                return ClassMethodNode(node, item, args, kwargs)

        return _MethodBinder()


class MultiOutputNode(DAGNode):
    def __init__(self, outputs: List[DAGNode]):
        self._outputs = outputs

    def execute(self, *input_args, _cache: Dict = None, **input_kwargs):
        cache = _cache if _cache is not None else {}
        return [
            o.execute(*input_args, _cache=cache, **input_kwargs)
            for o in self._outputs
        ]


def _resolve(v, input_args, input_kwargs, cache):
    if isinstance(v, InputNode):
        if len(input_args) == 1 and not input_kwargs:
            return input_args[0]
        return input_args
    if isinstance(v, InputAttributeNode):
        if v._key is not None:
            return input_kwargs[v._key]
        return input_args[v._index]
    if isinstance(v, DAGNode):
        key = id(v)
        if key not in cache:
            cache[key] = v.execute(*input_args, _cache=cache, **input_kwargs)
        return cache[key]
    return v


class InputAttributeNode(DAGNode):
    def __init__(self, index=None, key=None):
        self._index = index
        self._key = key


class _NotCompilable(Exception):
    pass


class DAGFuture:
    """Result handle from a channel-compiled DAG execution (reference:
    CompiledDAGRef — resolved with .get())."""

    def __init__(self, resolve):
        self._resolve = resolve
        self._done = False
        self._value = None

    def get(self, timeout: Optional[float] = 60.0):
        if not self._done:
            self._value = self._resolve(timeout)
            self._done = True
        return self._value


def _dag_loop(instance, specs):
    """Runs inside the actor (via __ray_apply__): ONE loop per actor
    executing all of that actor's bound nodes in topo order each
    iteration — reads input channels, runs the method, writes the
    result channel (reference: dag/compiled_dag_node.py exec loops,
    one loop per participating actor)."""
    from ray_amd.experimental.channel import Channel, ChannelReader

    from ray_amd.experimental import rdt as _rdt

    states = []
    for spec in specs:
        readers = {
            name: ChannelReader(Channel(path), slot)
            for name, (path, slot) in spec["reads"].items()
        }
        out = Channel(spec["out_path"]) if spec.get("out_path") else None
        # per-out-channel history of offloaded GPU tensors: the channel
        # ack protocol guarantees the reader finished computing with
        # iteration N-1's tensors by the time we WRITE iteration N+1
        # (it acked N before then), so a 2-deep history is the earliest
        # safe free point (accelerator-channel lifetime rule; reference:
        # torch_tensor_accelerator_channel.py buffer reuse)
        states.append((spec, readers, out, []))

    def render(t, vals):
        kind = t[0]
        if kind == "const":
            return t[1]
        if kind == "chan":
            return vals[t[1]]
        in_args, in_kwargs = vals["__input__"]
        if kind == "input":
            if len(in_args) == 1 and not in_kwargs:
                return in_args[0]
            return tuple(in_args)
        if kind == "input_idx":
            return in_args[t[1]]
        return in_kwargs[t[1]]  # input_key

    def _free_refs(refs):
        store = _rdt.get_gpu_object_store()
        for ref in refs:
            try:
                store.free(ref)
            except Exception:
                pass

    while True:
        stopping = False
        for spec, readers, out, gpu_hist in states:
            vals = {}
            stop = False
            err = None
            for name, rd in readers.items():
                tag, payload = rd.next_obj(timeout=None)
                if tag == "stop":
                    stop = True
                elif tag == "err":
                    err = payload
                else:
                    vals[name] = payload
            if stop:
                if out is not None:
                    out.write_obj(("stop", None))
                for refs in gpu_hist:
                    _free_refs(refs)
                gpu_hist.clear()
                stopping = True
                continue
            if err is not None:
                if out is not None:
                    out.write_obj(("err", err))
                continue
            try:
                args = [render(t, vals) for t in spec["args"]]
                kwargs = {
                    k: render(t, vals) for k, t in spec["kwargs"].items()
                }
                r = getattr(instance, spec["method"])(*args, **kwargs)
                if out is not None:
                    if _rdt.has_cuda_tensors(r):
                        # GPU edge: ship hipIpc refs, not host bytes
                        r, refs = _rdt.offload_tensors(r)
                        gpu_hist.append(refs)
                        if len(gpu_hist) > 2:
                            _free_refs(gpu_hist.pop(0))
                    out.write_obj(("data", r))
            except BaseException as e:  # noqa
                if out is not None:
                    out.write_obj(("err", e))
        if stopping:
            return "stopped"


class CompiledDAG:
    """Channel-driven compiled execution (reference:
    dag/compiled_dag_node.py): every actor in the DAG runs a persistent
    loop reading shm channels and writing its result channel, so a
    steady-state execution does zero task submissions. Falls back to
    per-call submission when the DAG contains plain function nodes.

    `execute()` on the channel path returns a DAGFuture (.get());
    on the fallback path it returns ObjectRef(s) as before."""

    def __init__(self, root: DAGNode):
        self._root = root
        self._channel_mode = False
        self._torn_down = False
        try:
            self._compile_channels()
            self._channel_mode = True
        except _NotCompilable:
            pass

    # ---------------- channel compilation ----------------

    def _compile_channels(self):
        import os

        from . import api as _api
        from .experimental.channel import Channel, ChannelReader, channel_path

        outputs = (
            self._root._outputs
            if isinstance(self._root, MultiOutputNode)
            else [self._root]
        )
        # topo-collect method nodes
        order: List[ClassMethodNode] = []
        seen = set()

        def visit(n):
            if not isinstance(n, ClassMethodNode):
                raise _NotCompilable
            if id(n) in seen:
                return
            for a in list(n._args) + list(n._kwargs.values()):
                if isinstance(a, ClassMethodNode):
                    visit(a)
                elif isinstance(a, (InputNode, InputAttributeNode)):
                    pass
                elif isinstance(a, DAGNode):
                    raise _NotCompilable
            seen.add(id(n))
            order.append(n)

        for o in outputs:
            visit(o)
        if not order:
            raise _NotCompilable

        dag_id = os.urandom(4).hex()
        node_idx = {id(n): i for i, n in enumerate(order)}

        # channel reader registration: chan key -> list of consumers
        consumers: Dict[str, list] = {"__input__": []}
        for i, n in enumerate(order):
            for a in list(n._args) + list(n._kwargs.values()):
                if isinstance(a, (InputNode, InputAttributeNode)):
                    if i not in consumers["__input__"]:
                        consumers["__input__"].append(i)
                elif isinstance(a, ClassMethodNode):
                    key = f"n{node_idx[id(a)]}"
                    consumers.setdefault(key, [])
                    if i not in consumers[key]:
                        consumers[key].append(i)
        for o in outputs:
            key = f"n{node_idx[id(o)]}"
            consumers.setdefault(key, []).append("driver")
        if not consumers["__input__"]:
            raise _NotCompilable

        def chan_file(key):
            return channel_path(f"dag{dag_id}_{key}")

        # create channels (driver is the creator for all of them)
        self._channels: Dict[str, Channel] = {}
        for key, cons in consumers.items():
            self._channels[key] = Channel(
                chan_file(key), num_readers=max(1, len(cons)), create=True
            )

        def template(a, i):
            if isinstance(a, InputNode):
                return ("input",)
            if isinstance(a, InputAttributeNode):
                if a._key is not None:
                    return ("input_key", a._key)
                return ("input_idx", a._index)
            if isinstance(a, ClassMethodNode):
                return ("chan", f"n{node_idx[id(a)]}")
            return ("const", a)

        # build specs, then launch ONE loop per distinct actor (multiple
        # bound methods of the same actor execute in topo order inside
        # that single loop — two loops would deadlock a 1-concurrency
        # actor)
        actor_specs: Dict[bytes, list] = {}
        actor_handles: Dict[bytes, Any] = {}
        for i, n in enumerate(order):
            reads = {}
            for a in list(n._args) + list(n._kwargs.values()):
                if isinstance(a, (InputNode, InputAttributeNode)):
                    reads["__input__"] = (
                        chan_file("__input__"),
                        consumers["__input__"].index(i),
                    )
                elif isinstance(a, ClassMethodNode):
                    key = f"n{node_idx[id(a)]}"
                    reads[key] = (chan_file(key), consumers[key].index(i))
            spec = {
                "method": n._method,
                "args": [template(a, i) for a in n._args],
                "kwargs": {k: template(v, i) for k, v in n._kwargs.items()},
                "reads": reads,
                "out_path": chan_file(f"n{i}"),
            }
            actor = n._actor_node._get_actor()
            actor_handles[actor._actor_id] = actor
            actor_specs.setdefault(actor._actor_id, []).append(spec)
        self._loop_refs = []
        self._actors = list(actor_handles.values())
        for aid, specs in actor_specs.items():
            ref = actor_handles[aid].__ray_apply__.remote(_dag_loop, specs)
            self._loop_refs.append(ref)

        # driver-side output readers
        self._out_readers = []
        for o in outputs:
            key = f"n{node_idx[id(o)]}"
            slot = consumers[key].index("driver")
            self._out_readers.append(
                ChannelReader(self._channels[key], slot)
            )
        self._multi = isinstance(self._root, MultiOutputNode)
        self._api = _api

    def execute(self, *args, **kwargs):
        if not self._channel_mode:
            return self._root.execute(*args, _cache={}, **kwargs)
        if self._torn_down:
            raise RuntimeError("CompiledDAG was torn down")
        self._channels["__input__"].write_obj(("data", (args, kwargs)))

        def resolve(timeout):
            outs = []
            first_err = None
            # drain EVERY output before raising, or a later execution
            # would read this one's stale error
            for rd in self._out_readers:
                tag, payload = rd.next_obj(timeout=timeout)
                if tag == "err" and first_err is None:
                    first_err = payload
                outs.append(payload)
            if first_err is not None:
                raise first_err
            return outs if self._multi else outs[0]

        return DAGFuture(resolve)

    def teardown(self):
        if not self._channel_mode or self._torn_down:
            return
        self._torn_down = True
        try:
            self._channels["__input__"].write_obj(("stop", None))
            self._api.get(self._loop_refs, timeout=30)
        except Exception:
            pass
        for rd in self._out_readers:
            # drain the stop marker so channel files can be reclaimed
            try:
                rd.next_obj(timeout=1.0)
            except Exception:
                pass
        for ch in self._channels.values():
            ch.close()
