"""tfdbg-style debugging (compact analog of the reference python/debug/
package): tensor watches, value dumps, and NaN/Inf detection around
Session.run — implemented by adding watched tensors as extra fetches rather
than injecting DebugIdentity nodes into the partition graphs."""
import fnmatch
import os
import re
import time

import numpy as np

from simple_tensorflow_amd.python.framework import ops


def has_inf_or_nan(datum, tensor_value):
    """Reference debug_data.has_inf_or_nan filter."""
    v = np.asarray(tensor_value)
    if v.dtype == object or v.dtype.kind in ('S', 'U'):
        return False
    return bool(np.isnan(v).any() or np.isinf(v).any())


class DebugTensorDatum(object):
    def __init__(self, node_name, output_slot, value, wall_time):
        self.node_name = node_name
        self.output_slot = output_slot
        self.tensor_name = '%s:%d' % (node_name, output_slot)
        self.value = value
        self.wall_time = wall_time

    @property
    def watch_key(self):
        return self.tensor_name + ':DebugIdentity'


class DebugDumpDir(object):
    """Loads dump .npy files written by DumpingDebugWrapperSession."""

    def __init__(self, dump_root):
        self._data = []
        for fn in sorted(os.listdir(dump_root)):
            if not fn.endswith('.npy'):
                continue
            m = re.match(r'(.*)__(\d+)__(\d+)\.npy$', fn)
            if not m:
                continue
            name = m.group(1).replace('~', '/')
            self._data.append(DebugTensorDatum(
                name, int(m.group(2)),
                np.load(os.path.join(dump_root, fn), allow_pickle=False),
                int(m.group(3))))

    @property
    def dumped_tensor_data(self):
        return list(self._data)

    def get_tensors(self, node_name, output_slot=0,
                    debug_op='DebugIdentity'):
        return [d.value for d in self._data
                if d.node_name == node_name and
                d.output_slot == output_slot]

    def find(self, predicate):
        return [d for d in self._data if predicate(d, d.value)]


class DumpingDebugWrapperSession(object):
    """Wraps a Session; every run() also evaluates watched tensors and dumps
    them under dump_root/run_<n>/ (reference
    debug/wrappers/dumping_wrapper.py)."""

    def __init__(self, sess, dump_root, watch_fn=None,
                 node_name_regex=None):
        self._sess = sess
        self._dump_root = dump_root
        self._watch_fn = watch_fn
        self._node_name_regex = re.compile(node_name_regex) \
            if node_name_regex else None
        self._run_counter = 0
        os.makedirs(dump_root, exist_ok=True)

    @property
    def graph(self):
        return self._sess.graph

    def _watched_tensors(self):
        g = self._sess.graph
        out = []
        for op in g._node_list:
            if op.type in ('Placeholder', 'NoOp', 'Assert'):
                continue
            if self._node_name_regex and \
                    not self._node_name_regex.match(op.name):
                continue
            for t in op.outputs:
                if t.dtype.name in ('float32', 'float64', 'bfloat16',
                                    'int32', 'int64'):
                    out.append(t)
        return out

    def run(self, fetches, feed_dict=None, options=None, run_metadata=None):
        watched = self._watched_tensors()
        run_dir = os.path.join(self._dump_root,
                               'run_%d' % self._run_counter)
        self._run_counter += 1
        os.makedirs(run_dir, exist_ok=True)
        result = self._sess.run(fetches, feed_dict=feed_dict)
        wall = int(time.time() * 1e6)
        for t in watched:
            try:
                v = self._sess.run(t, feed_dict=feed_dict)
            except Exception:
                continue
            fn = '%s__%d__%d.npy' % (t.op.name.replace('/', '~'),
                                     t.value_index, wall)
            np.save(os.path.join(run_dir, fn), np.asarray(v),
                    allow_pickle=False)
        return result

    def latest_dump_dir(self):
        return os.path.join(self._dump_root,
                            'run_%d' % (self._run_counter - 1))

    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False


def watch_graph(run_options, graph, debug_ops=None, debug_urls=None,
                node_name_regex_whitelist=None, op_type_regex_whitelist=None):
    """Reference debug_utils.watch_graph signature; records the watch spec
    on run_options for wrapper sessions to use."""
    run_options.debug_node_name_regex = node_name_regex_whitelist
    run_options.debug_op_type_regex = op_type_regex_whitelist
    return run_options
