"""Session-level graph optimization: CSE + constant folding
(csrc/graph/optimizer.cc; reference core/graph/optimizer_cse.cc +
common_runtime/constant_folding.cc capability analogs). Observed through
FULL_TRACE step stats: optimized-away nodes never execute."""
import numpy as np

import simple_tensorflow_amd as tf


def setup_function(_):
    tf.reset_default_graph()


def _executed_nodes(fetch, feed=None):
    opts = tf.RunOptions(trace_level=tf.RunOptions.FULL_TRACE)
    md = tf.RunMetadata()
    with tf.Session() as s:
        val = s.run(fetch, feed_dict=feed, options=opts, run_metadata=md)
    names = {ns.node_name for ds in md.step_stats.dev_stats
             for ns in ds.node_stats}
    return val, names


def test_constant_folding():
    a = tf.constant(np.full((4, 4), 2.0, np.float32), name='ca')
    b = tf.constant(np.full((4, 4), 3.0, np.float32), name='cb')
    m = tf.matmul(a, b, name='foldme')
    p = tf.placeholder(tf.float32, [4, 4], name='p')
    out = tf.add(m, p, name='out')
    val, names = _executed_nodes(out, {p: np.zeros((4, 4), np.float32)})
    np.testing.assert_allclose(val, np.full((4, 4), 24.0), rtol=1e-6)
    assert 'foldme' not in names  # folded to a Const before execution


def test_cse_merges_duplicates():
    p = tf.placeholder(tf.float32, [8], name='p')
    a = tf.multiply(p, p, name='sq1')
    b = tf.multiply(p, p, name='sq2')
    out = tf.add(a, b, name='out')
    x = np.random.randn(8).astype(np.float32)
    val, names = _executed_nodes(out, {p: x})
    np.testing.assert_allclose(val, 2 * x * x, rtol=1e-5)
    # exactly one of the two identical Muls executed
    assert ('sq1' in names) != ('sq2' in names)


def test_fetched_node_preserved():
    a = tf.constant(np.arange(6, dtype=np.float32), name='a')
    m = tf.multiply(a, a, name='fetched')
    val, names = _executed_nodes(m)
    np.testing.assert_allclose(val, np.arange(6) ** 2)


def test_variables_not_folded():
    v = tf.Variable(np.full((3,), 5.0, np.float32), name='v')
    out = tf.multiply(v, tf.constant(2.0), name='double')
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        np.testing.assert_allclose(s.run(out), [10.0, 10.0, 10.0])
        s.run(tf.assign(v, np.full((3,), 7.0, np.float32)))
        np.testing.assert_allclose(s.run(out), [14.0, 14.0, 14.0])


def test_random_not_merged():
    # Two RandomUniform nodes with identical attrs must stay distinct.
    a = tf.random_uniform([1000], seed=None, name='r1')
    b = tf.random_uniform([1000], seed=None, name='r2')
    d = tf.reduce_max(tf.abs(a - b))
    with tf.Session() as s:
        assert s.run(d) > 0.0


def test_conv_dx_add_fusion_cpu():
    """Gradient aggregation folds AddN(Conv2DBackpropInput, side) into
    Conv2DBackpropInputAdd (residual-block pattern); numerics must match
    the plain composition."""
    import os
    import numpy as np
    tf.reset_default_graph()
    rng = np.random.RandomState(3)
    xv = rng.randn(2, 6, 6, 5).astype(np.float32)
    wv = rng.randn(3, 3, 5, 7).astype(np.float32)
    bv = rng.randn(2, 6, 6, 5).astype(np.float32)
    x = tf.constant(xv)
    w = tf.constant(wv)
    b = tf.constant(bv)
    y = tf.nn.conv2d(x, w, [1, 1, 1, 1], 'SAME')
    loss = tf.reduce_sum(y * 2.0) + tf.reduce_sum(x * b)
    g, = tf.gradients(loss, [x])
    assert g.op.type == 'Conv2DBackpropInputAdd'
    with tf.Session() as s:
        gv = s.run(g)
    os.environ['STF_NO_CONV_DX_FUSE'] = '1'
    try:
        tf.reset_default_graph()
        x = tf.constant(xv)
        w = tf.constant(wv)
        b = tf.constant(bv)
        y = tf.nn.conv2d(x, w, [1, 1, 1, 1], 'SAME')
        loss = tf.reduce_sum(y * 2.0) + tf.reduce_sum(x * b)
        g2, = tf.gradients(loss, [x])
        assert g2.op.type == 'AddN'
        with tf.Session() as s:
            gv2 = s.run(g2)
    finally:
        del os.environ['STF_NO_CONV_DX_FUSE']
    np.testing.assert_allclose(gv, gv2, rtol=1e-5, atol=1e-5)


def test_conv_dx_add_fusion_skips_shared_grad():
    """When the conv-backprop partial is consumed elsewhere the rewrite
    must not fire (it would duplicate the GEMM)."""
    import numpy as np
    tf.reset_default_graph()
    x = tf.constant(np.ones((1, 4, 4, 3), np.float32))
    w = tf.constant(np.ones((1, 1, 3, 3), np.float32))
    b = tf.constant(np.ones((1, 4, 4, 3), np.float32))
    y = tf.nn.conv2d(x, w, [1, 1, 1, 1], 'SAME')
    # y + y: the Add grad propagates the SAME incoming tensor to both
    # slots; downstream aggregation sees a list with a repeated tensor
    z = y + y
    loss = tf.reduce_sum(z) + tf.reduce_sum(x * b)
    g, = tf.gradients(loss, [x])
    with tf.Session() as s:
        gv = s.run(g)
    np.testing.assert_allclose(gv, np.full((1, 4, 4, 3), 7.0), rtol=1e-5)
