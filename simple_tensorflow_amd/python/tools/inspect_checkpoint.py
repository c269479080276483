"""inspect_checkpoint: print tensor names/shapes (and values) in a V2
checkpoint (reference python/tools/inspect_checkpoint.py over
checkpoint_reader.cc — here over _core.list_checkpoint + RestoreV2)."""
import argparse

from simple_tensorflow_amd import _core
from simple_tensorflow_amd.python.framework import dtypes


class CheckpointReader(object):
    """NewCheckpointReader-style interface."""

    def __init__(self, prefix):
        self._prefix = prefix
        self._entries = {name: (dtype, shape)
                         for name, dtype, shape in
                         _core.list_checkpoint(prefix)}

    def get_variable_to_shape_map(self):
        return {n: list(s) for n, (d, s) in self._entries.items()}

    def get_variable_to_dtype_map(self):
        return {n: dtypes.as_dtype(d) for n, (d, s) in
                self._entries.items()}

    def has_tensor(self, name):
        return name in self._entries

    def get_tensor(self, name):
        import simple_tensorflow_amd as tf
        from simple_tensorflow_amd.python.framework.ops import apply_op
        dtype, shape = self._entries[name]
        g = tf.Graph()
        with g.as_default():
            fn = tf.placeholder(tf.string, [], name='ckpt_fn')
            names = tf.constant([name.encode()])
            slices = tf.constant([b''])
            t = apply_op('RestoreV2', fn, names, slices,
                         dtypes=[dtypes.as_dtype(dtype)])
            with tf.Session(graph=g) as s:
                return s.run(t, {fn: self._prefix})


def NewCheckpointReader(prefix):
    return CheckpointReader(prefix)


def print_tensors_in_checkpoint_file(file_name, tensor_name=None,
                                     all_tensors=False):
    reader = CheckpointReader(file_name)
    if tensor_name and not all_tensors:
        print('tensor_name: ', tensor_name)
        print(reader.get_tensor(tensor_name))
        return
    shape_map = reader.get_variable_to_shape_map()
    for name in sorted(shape_map):
        print('tensor_name: ', name, shape_map[name])
        if all_tensors:
            print(reader.get_tensor(name))


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--file_name', required=True)
    p.add_argument('--tensor_name', default='')
    p.add_argument('--all_tensors', action='store_true')
    a = p.parse_args()
    print_tensors_in_checkpoint_file(a.file_name, a.tensor_name or None,
                                     a.all_tensors)


if __name__ == '__main__':
    main()
