"""tf.train.Saver — checkpoint save/restore through in-graph SaveV2/RestoreV2
ops over the tensor-bundle format (analog of reference
python/training/saver.py:872 + BaseSaverBuilder:82; kernels in
csrc/kernels/cpu_state_io.cc keep the on-disk format compatible)."""
import os
import re
import time

from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import apply_op
from simple_tensorflow_amd.python.ops import array_ops, control_flow_ops, state_ops, variables


class Saver(object):
    def __init__(self, var_list=None, max_to_keep=5, name='save',
                 keep_checkpoint_every_n_hours=10000.0, sharded=False,
                 restore_sequentially=False, **kw):
        g = ops.get_default_graph()
        if var_list is None:
            var_list = variables.global_variables()
        if isinstance(var_list, dict):
            self._names = list(var_list.keys())
            self._vars = [var_list[k] for k in self._names]
        else:
            self._vars = list(var_list)
            self._names = [v.name.split(':')[0] for v in self._vars]
        self._max_to_keep = max_to_keep
        self._kept = []

        with g.name_scope(name):
            self._filename = array_ops.placeholder(dtypes.string, [],
                                                   name='filename')
            names_c = ops.constant([n.encode() for n in self._names])
            slices_c = ops.constant([b''] * len(self._names))
            tensors = [v.value() if hasattr(v, 'value') else v
                       for v in self._vars]
            self._save_op = apply_op('SaveV2', self._filename, names_c,
                                     slices_c, tensors, name='SaveV2')
            dts = [t.dtype for t in tensors]
            restored = apply_op('RestoreV2', self._filename, names_c,
                                slices_c, dtypes=dts, name='RestoreV2')
            if not isinstance(restored, (list, tuple)):
                restored = [restored]
            assigns = []
            for v, r, t in zip(self._vars, restored, tensors):
                r.set_shape(t._shape)
                ref = v._as_graph_element() if hasattr(v, '_as_graph_element') \
                    else v
                assigns.append(state_ops.assign(ref, r,
                                                validate_shape=False).op)
            self._restore_op = control_flow_ops.group(*assigns,
                                                      name='restore_all')

    @classmethod
    def _from_imported(cls, filename_tensor, save_op, restore_op,
                       max_to_keep=5):
        """Wrap save/restore nodes recovered by import_meta_graph."""
        s = cls.__new__(cls)
        s._filename = filename_tensor
        s._save_op = save_op
        s._restore_op = restore_op
        s._max_to_keep = max_to_keep
        s._kept = []
        s._names = []
        s._vars = []
        return s

    def export_meta_graph(self, filename=None, collection_list=None):
        from simple_tensorflow_amd.python.framework import meta_graph
        return meta_graph.export_meta_graph(
            filename=filename, saver=self, collection_list=collection_list)

    def save(self, sess, save_path, global_step=None,
             latest_filename='checkpoint', write_meta_graph=True,
             meta_graph_suffix='meta'):
        if global_step is not None:
            if hasattr(global_step, 'eval') or isinstance(global_step,
                                                          ops.Tensor):
                from simple_tensorflow_amd.python.client import session as sm
                step = int(sess.run(global_step if isinstance(
                    global_step, ops.Tensor) else global_step.value()))
            else:
                step = int(global_step)
            path = '%s-%d' % (save_path, step)
        else:
            path = save_path
        sess.run(self._save_op, feed_dict={self._filename: path})
        self._record_checkpoint(path, latest_filename)
        if write_meta_graph:
            try:
                self.export_meta_graph(path + '.' + meta_graph_suffix)
            except Exception:
                pass  # meta export must not fail the checkpoint itself
        return path

    def restore(self, sess, save_path):
        sess.run(self._restore_op, feed_dict={self._filename: save_path})

    def last_checkpoints(self):
        return list(self._kept)

    def _record_checkpoint(self, path, latest_filename):
        d = os.path.dirname(path) or '.'
        self._kept.append(path)
        while len(self._kept) > self._max_to_keep:
            old = self._kept.pop(0)
            for suffix in ('.index', '.data-00000-of-00001'):
                try:
                    os.remove(old + suffix)
                except OSError:
                    pass
        update_checkpoint_state(d, path, self._kept, latest_filename)


def update_checkpoint_state(save_dir, model_checkpoint_path,
                            all_model_checkpoint_paths=None,
                            latest_filename='checkpoint'):
    """Writes the text-format CheckpointState file (reference
    python/training/saver.py generate_checkpoint_state_proto)."""
    lines = ['model_checkpoint_path: "%s"' % model_checkpoint_path]
    for p in (all_model_checkpoint_paths or [model_checkpoint_path]):
        lines.append('all_model_checkpoint_paths: "%s"' % p)
    with open(os.path.join(save_dir, latest_filename), 'w') as f:
        f.write('\n'.join(lines) + '\n')


def get_checkpoint_state(checkpoint_dir, latest_filename='checkpoint'):
    path = os.path.join(checkpoint_dir, latest_filename)
    if not os.path.exists(path):
        return None

    class State(object):
        model_checkpoint_path = None
        all_model_checkpoint_paths = []
    st = State()
    st.all_model_checkpoint_paths = []
    with open(path) as f:
        for line in f:
            m = re.match(r'model_checkpoint_path: "(.*)"', line.strip())
            if m:
                st.model_checkpoint_path = m.group(1)
            m = re.match(r'all_model_checkpoint_paths: "(.*)"', line.strip())
            if m:
                st.all_model_checkpoint_paths.append(m.group(1))
    return st


def latest_checkpoint(checkpoint_dir, latest_filename='checkpoint'):
    st = get_checkpoint_state(checkpoint_dir, latest_filename)
    return st.model_checkpoint_path if st else None


def checkpoint_exists(path):
    return os.path.exists(path + '.index')
