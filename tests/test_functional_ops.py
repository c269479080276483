"""map_fn / foldl / foldr / scan (python/ops/functional_ops.py; reference
functional_ops.py analog) + full TensorArray gradients (gather/scatter/
read/write through the shadow gradient array)."""
import numpy as np

import simple_tensorflow_amd as tf


def setup_function(_):
    tf.reset_default_graph()


def test_map_fn():
    x = np.arange(6, dtype=np.float32)
    with tf.Session() as s:
        out = s.run(tf.map_fn(lambda e: e * e + 1.0, tf.constant(x)))
    np.testing.assert_allclose(out, x * x + 1.0)


def test_map_fn_matrix_rows():
    x = np.random.RandomState(0).randn(4, 3).astype(np.float32)
    with tf.Session() as s:
        out = s.run(tf.map_fn(lambda r: tf.reduce_sum(r) * r,
                              tf.constant(x)))
    np.testing.assert_allclose(out, x.sum(1, keepdims=True) * x, rtol=1e-5)


def test_foldl_foldr():
    x = np.arange(1, 6, dtype=np.float32)
    with tf.Session() as s:
        l = s.run(tf.foldl(lambda a, e: a * e, tf.constant(x)))
        r = s.run(tf.foldr(lambda e, a: a - e, tf.constant(x),
                           tf.constant(0.0)))
    np.testing.assert_allclose(l, np.prod(x))
    # foldr: 0 - 5 - 4 - 3 - 2 - 1
    np.testing.assert_allclose(r, -x.sum())


def test_scan_cumsum():
    x = np.random.RandomState(1).randn(7).astype(np.float32)
    with tf.Session() as s:
        out = s.run(tf.scan(lambda a, e: a + e, tf.constant(x)))
    np.testing.assert_allclose(out, np.cumsum(x), rtol=1e-5)


def test_map_fn_gradient():
    with tf.Session() as s:
        ph = tf.placeholder(tf.float32, [4])
        y = tf.reduce_sum(tf.map_fn(lambda e: e * e, ph))
        g = s.run(tf.gradients(y, [ph])[0],
                  {ph: np.array([1., 2., 3., 4.], np.float32)})
    np.testing.assert_allclose(g, [2., 4., 6., 8.])


def test_scan_gradient_numeric():
    x0 = np.array([1., 2., 3., 4., 5.], np.float32)
    with tf.Session() as s:
        ph = tf.placeholder(tf.float32, [5])
        loss = tf.reduce_sum(tf.scan(lambda a, e: a * e, ph,
                                     tf.constant(1.0)))
        gv = s.run(tf.gradients(loss, [ph])[0], {ph: x0})
        num = np.zeros(5)
        for k in range(5):
            e = 1e-3
            xp, xm = x0.copy(), x0.copy()
            xp[k] += e
            xm[k] -= e
            num[k] = (np.cumprod(xp).sum() - np.cumprod(xm).sum()) / (2 * e)
    np.testing.assert_allclose(gv, num, rtol=1e-2)


def test_dynamic_rnn_output_gradients():
    """The stacked dynamic_rnn outputs (TensorArrayGather) are now
    differentiable — loss over ALL timesteps, not just the final state."""
    np.random.seed(3)
    x = np.random.randn(2, 5, 3).astype(np.float32)
    cell = tf.nn.rnn_cell.BasicRNNCell(4)
    out, state = tf.nn.dynamic_rnn(cell, tf.constant(x), dtype=tf.float32)
    loss = tf.reduce_sum(out * out)
    opt = tf.train.GradientDescentOptimizer(0.05)
    train = opt.minimize(loss)
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        l0 = s.run(loss)
        for _ in range(10):
            s.run(train)
        l1 = s.run(loss)
    assert np.isfinite(l1) and l1 < l0


def test_tensor_array_duplicate_read_grads_accumulate():
    """A TA slot read twice contributes both read-gradients: the shadow
    gradient array must ACCUMULATE duplicate-index writes (reference
    tensor_array.h multiple_writes_aggregate), not keep the last one."""
    tf.reset_default_graph()
    v = tf.constant(np.array([1.0, 2.0], np.float32))
    ta = tf.TensorArray(tf.float32, size=1)
    ta = ta.write(0, v)
    a = ta.read(0)
    b = ta.read(0)
    loss = tf.reduce_sum(a * 3.0) + tf.reduce_sum(b * 5.0)
    g, = tf.gradients(loss, [v])
    with tf.Session() as s:
        gv = s.run(g)
    np.testing.assert_allclose(gv, [8.0, 8.0])
