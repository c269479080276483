"""tf.Session — client for the C++ DirectSession.

Analog of the reference's python/client/session.py (BaseSession.run:660),
speaking to csrc/runtime/session.cc through the pybind module.
"""
import threading
import time

import numpy as np

from simple_tensorflow_amd import _core
from simple_tensorflow_amd.python.framework import dtypes, errors, ops
from simple_tensorflow_amd.python.lib import monitoring

# Framework metrics (reference core/lib/monitoring instrumentation points).
_run_counter = monitoring.Counter('/stf/session/runs',
                                  'Number of Session.run calls')
_run_time_ms = monitoring.Sampler(
    '/stf/session/run_time_ms', monitoring.exponential_buckets(0.1, 4, 12),
    'Session.run wall time (ms)')

_default_session_stack = threading.local()


def get_default_session():
    stack = getattr(_default_session_stack, 'stack', [])
    if not stack:
        raise RuntimeError('No default session')
    return stack[-1]


class Session(object):
    def __init__(self, target='', graph=None, config=None):
        self._graph = graph if graph is not None else ops.get_default_graph()
        cpu_only = False
        if config is not None and isinstance(config, dict):
            cpu_only = config.get('device_count', {}).get('GPU', 1) == 0
        if target.startswith('grpc://'):
            from simple_tensorflow_amd.python.training import server_lib
            self._core = server_lib.GrpcRemoteCore(target)
        else:
            self._core = _core.Session(cpu_only)
        self._created = False
        self._serialized_nodes = 0
        self._operation_timeout_ms = 0
        if config is not None and isinstance(config, dict):
            self._operation_timeout_ms = int(
                config.get('operation_timeout_in_ms', 0))
        self._lock = threading.Lock()

    def partial_run_setup(self, fetches, feeds=None):
        """Declare the fetches/feeds of an incremental run (reference
        Session.partial_run_setup): executors start now; partial_run() feeds
        and fetches subsets until every declared fetch is consumed."""
        self._sync_graph()
        if not isinstance(fetches, (list, tuple)):
            fetches = [fetches]
        feeds = feeds or []
        if not isinstance(feeds, (list, tuple)):
            feeds = [feeds]
        fetch_names = [_as_fetchable(f, self._graph).name for f in fetches]
        feed_names = [_as_fetchable(f, self._graph).name for f in feeds]
        return self._core.partial_run_setup(feed_names, fetch_names, [])

    def partial_run(self, handle, fetches, feed_dict=None):
        single = not isinstance(fetches, (list, tuple))
        if single:
            fetches = [fetches]
        fetch_names = [_as_fetchable(f, self._graph).name for f in fetches]
        feeds = {}
        if feed_dict:
            for k, v in feed_dict.items():
                t = _as_fetchable(k, self._graph)
                feeds[t.name] = _convert_feed(t, v)
        try:
            results = self._core.partial_run(handle, feeds, fetch_names)
        except RuntimeError as e:
            errors.raise_from_message(str(e))
        out = list(results)
        return out[0] if single else out

    @staticmethod
    def reset(target='', containers=None, config=None):
        """Resets resource containers (reference Session.reset / TF_Reset):
        drops every stateful kernel (variables, queues, lookup tables) of all
        live in-process sessions; they see fresh uninitialized state."""
        _core.Session.reset_all()

    @property
    def graph(self):
        return self._graph

    def _sync_graph(self):
        g = self._graph
        with self._lock:
            if not self._created or g._mutated_after_serialize:
                self._core.create(g.as_graph_def(0))
                self._created = True
                g._mutated_after_serialize = False
                self._serialized_nodes = len(g._node_list)
            elif len(g._node_list) > self._serialized_nodes:
                self._core.extend(g.as_graph_def(self._serialized_nodes))
                self._serialized_nodes = len(g._node_list)

    def run(self, fetches, feed_dict=None, options=None, run_metadata=None):
        _run_counter.get_cell().increment()
        _t0 = time.perf_counter()
        try:
            return self._run(fetches, feed_dict, options, run_metadata)
        finally:
            _run_time_ms.get_cell().add((time.perf_counter() - _t0) * 1e3)

    def _run(self, fetches, feed_dict=None, options=None, run_metadata=None):
        self._sync_graph()
        flat, restore = _flatten_fetches(fetches)
        fetch_names = []
        targets = []
        fetch_slots = []  # index into results for each flat fetch, or None
        for f in flat:
            if isinstance(f, ops.Operation):
                targets.append(f.name)
                fetch_slots.append(None)
            else:
                t = _as_fetchable(f, self._graph)
                if isinstance(t, ops.Operation):
                    targets.append(t.name)
                    fetch_slots.append(None)
                else:
                    fetch_slots.append(len(fetch_names))
                    fetch_names.append(t.name)

        feeds = {}
        if feed_dict:
            for k, v in feed_dict.items():
                t = _as_fetchable(k, self._graph)
                feeds[t.name] = _convert_feed(t, v)

        want_stats = (options is not None and
                      getattr(options, 'trace_level', 0) and
                      run_metadata is not None)
        timeout_ms = int(getattr(options, 'timeout_in_ms', 0) or 0)
        if not timeout_ms:
            timeout_ms = self._operation_timeout_ms
        try:
            if want_stats or timeout_ms:
                results = self._core.run(feeds, fetch_names, targets,
                                         want_stats, timeout_ms)
            else:
                results = self._core.run(feeds, fetch_names, targets)
        except RuntimeError as e:
            errors.raise_from_message(str(e))
        except TypeError:
            # remote (grpc) cores take no collect_stats flag
            results = self._core.run(feeds, fetch_names, targets)
            want_stats = False
        if want_stats:
            from simple_tensorflow_amd.python.client import timeline as tl
            run_metadata.step_stats = tl.StepStats(self._core.last_stats())
        out = []
        for f, slot in zip(flat, fetch_slots):
            if slot is None:
                out.append(None)
            else:
                val = results[slot]
                t = f if isinstance(f, ops.Tensor) else None
                if t is not None and t._shape is not None and \
                        isinstance(val, np.ndarray) and val.ndim == 0:
                    pass
                out.append(val)
        return restore(out)

    def close(self):
        pass

    def __enter__(self):
        stack = getattr(_default_session_stack, 'stack', None)
        if stack is None:
            _default_session_stack.stack = stack = []
        stack.append(self)
        return self

    def __exit__(self, *exc):
        _default_session_stack.stack.pop()
        return False

    def as_default(self):
        return self

    def num_gpus(self):
        return self._core.num_gpus()

    def sync(self):
        """Block until all enqueued device work completes."""
        self._core.sync()


class InteractiveSession(Session):
    def __init__(self, *a, **kw):
        super().__init__(*a, **kw)
        stack = getattr(_default_session_stack, 'stack', None)
        if stack is None:
            _default_session_stack.stack = stack = []
        stack.append(self)


def _as_fetchable(f, graph):
    if isinstance(f, (ops.Tensor, ops.Operation)):
        return f
    if isinstance(f, str):
        if ':' in f:
            return graph.get_tensor_by_name(f)
        return graph.get_operation_by_name(f)
    if hasattr(f, '_as_graph_element'):
        return f._as_graph_element()
    raise TypeError('Cannot fetch %r' % (f,))


def _flatten_fetches(fetches):
    if isinstance(fetches, (list, tuple)):
        items = list(fetches)
        def restore(vals):
            return type(fetches)(vals) if isinstance(fetches, tuple) else vals
        return items, restore
    if isinstance(fetches, dict):
        keys = list(fetches.keys())
        items = [fetches[k] for k in keys]
        def restore(vals):
            return dict(zip(keys, vals))
        return items, restore
    return [fetches], lambda vals: vals[0]


def _check_feed_shape(t, arr):
    # Reference BaseSession._run raises ValueError when a fed array is
    # incompatible with the placeholder's declared static shape; silent
    # acceptance invalidates static-shape-based gradient fast paths.
    shape = getattr(t, '_shape', None)
    if shape is None:
        return
    if len(arr.shape) != len(shape) or any(
            d is not None and d != a for d, a in zip(shape, arr.shape)):
        raise ValueError(
            'Cannot feed value of shape %r for Tensor %r, which has shape %r'
            % (tuple(arr.shape), t.name, tuple(shape)))


def _convert_feed(t, v):
    if t.dtype is dtypes.string:
        if isinstance(v, str):
            return v.encode()
        if isinstance(v, bytes):
            return v
        raise TypeError('string feed must be str/bytes')
    np_dt = t.dtype.as_numpy_dtype
    arr = np.asarray(v)
    _check_feed_shape(t, arr)
    if t.dtype is dtypes.bfloat16:
        arr32 = arr.astype(np.float32)
        return ops._f32_to_bf16(arr32)
    if arr.dtype != np_dt:
        arr = arr.astype(np_dt)
    return np.ascontiguousarray(arr)
