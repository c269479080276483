"""Set ops (csrc/kernels/cpu_sets.cc; reference core/ops/set_ops.cc)."""
import numpy as np

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.ops import sets_impl


def setup_function(_):
    tf.reset_default_graph()


def _dense(sp):
    with tf.Session() as s:
        return s.run([sp.indices, sp.values, sp.dense_shape])


def test_intersection_union_difference():
    a = np.array([[1, 2, 3], [4, 5, 6]], np.int32)
    b = np.array([[2, 3, 7], [9, 5, 5]], np.int32)
    i, v, sh = _dense(sets_impl.set_intersection(a, b))
    assert v.tolist() == [2, 3, 5]
    assert i.tolist() == [[0, 0], [0, 1], [1, 0]]
    _, vu, shu = _dense(sets_impl.set_union(a, b))
    assert sorted(vu.tolist()) == [1, 2, 3, 4, 5, 6, 7, 9]
    assert shu.tolist() == [2, 4]
    _, vd, _ = _dense(sets_impl.set_difference(a, b))
    assert vd.tolist() == [1, 4, 6]
    _, vd2, _ = _dense(sets_impl.set_difference(a, b, aminusb=False))
    assert vd2.tolist() == [7, 9]


def test_set_size():
    a = np.array([[1, 2, 2], [3, 3, 3]], np.int32)
    b = np.array([[2, 9, 9], [1, 1, 1]], np.int32)
    un = sets_impl.set_union(a, b)
    sz = sets_impl.set_size(un)
    with tf.Session() as s:
        got = s.run(sz)
    assert got.tolist() == [3, 2]  # {1,2,9}, {1,3}
