"""tfdbg-style debugging (reference python/debug analogs)."""
import numpy as np

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python import debug as tf_debug


def setup_function(_):
    tf.reset_default_graph()


def test_dump_and_find_nan(tmp_path):
    x = tf.constant([1.0, 0.0], name='x')
    y = tf.divide(tf.constant([1.0, 1.0]), x, name='div')  # inf at idx 1
    z = tf.multiply(y, tf.constant(0.0), name='mul')       # nan
    sess = tf.Session()
    with sess:
        dbg = tf_debug.DumpingDebugWrapperSession(sess, str(tmp_path))
        out = dbg.run(z)
        dump = tf_debug.DebugDumpDir(dbg.latest_dump_dir())
        bad = dump.find(tf_debug.has_inf_or_nan)
    bad_nodes = {d.node_name for d in bad}
    assert 'div' in bad_nodes and 'mul' in bad_nodes
    assert 'x' not in bad_nodes
    vals = dump.get_tensors('x')
    np.testing.assert_allclose(vals[0], [1.0, 0.0])


def test_regex_filtering(tmp_path):
    tf.constant([2.0], name='keepme')
    tf.constant([3.0], name='dropme')
    sess = tf.Session()
    with sess:
        dbg = tf_debug.DumpingDebugWrapperSession(
            sess, str(tmp_path), node_name_regex='keep.*')
        dbg.run(tf.get_default_graph().get_tensor_by_name('keepme:0'))
        dump = tf_debug.DebugDumpDir(dbg.latest_dump_dir())
    names = {d.node_name for d in dump.dumped_tensor_data}
    assert names == {'keepme'}
