"""Core framework tests: session, feeds, variables, gradients, control flow.

Mirrors the reference's python kernel-test strategy: numpy
reference comparison on CPU.
"""
import numpy as np
import pytest

import simple_tensorflow_amd as tf


@pytest.fixture(autouse=True)
def fresh_graph():
    tf.reset_default_graph()
    yield


def test_constant_roundtrip():
    with tf.Session() as s:
        c = tf.constant([[1.0, 2.0], [3.0, 4.0]])
        np.testing.assert_allclose(s.run(c), [[1, 2], [3, 4]])


def test_matmul_vs_numpy():
    a = np.random.randn(7, 5).astype(np.float32)
    b = np.random.randn(5, 3).astype(np.float32)
    with tf.Session() as s:
        out = s.run(tf.matmul(tf.constant(a), tf.constant(b)))
    np.testing.assert_allclose(out, a @ b, rtol=1e-5, atol=1e-6)


def test_matmul_transpose():
    a = np.random.randn(5, 7).astype(np.float32)
    b = np.random.randn(3, 5).astype(np.float32)
    with tf.Session() as s:
        out = s.run(tf.matmul(tf.constant(a), tf.constant(b),
                              transpose_a=True, transpose_b=True))
    np.testing.assert_allclose(out, a.T @ b.T, rtol=1e-5, atol=1e-6)


def test_feed_fetch():
    with tf.Session() as s:
        x = tf.placeholder(tf.float32, [3])
        y = x * 2.0 + 1.0
        out = s.run(y, feed_dict={x: np.array([1, 2, 3], np.float32)})
        np.testing.assert_allclose(out, [3, 5, 7])


def test_broadcasting():
    a = np.random.randn(4, 1, 3).astype(np.float32)
    b = np.random.randn(5, 1).astype(np.float32)
    with tf.Session() as s:
        out = s.run(tf.add(tf.constant(a), tf.constant(b)))
    np.testing.assert_allclose(out, a + b, rtol=1e-6)


def test_reductions():
    x = np.random.randn(4, 5, 6).astype(np.float32)
    with tf.Session() as s:
        t = tf.constant(x)
        np.testing.assert_allclose(s.run(tf.reduce_sum(t)), x.sum(), rtol=1e-4)
        np.testing.assert_allclose(s.run(tf.reduce_mean(t, 1)), x.mean(1),
                                   rtol=1e-5, atol=1e-5)
        np.testing.assert_allclose(s.run(tf.reduce_max(t, [0, 2])),
                                   x.max((0, 2)), rtol=1e-6)


def test_variables_and_assign():
    with tf.Session() as s:
        v = tf.Variable(np.arange(6, dtype=np.float32).reshape(2, 3))
        s.run(tf.global_variables_initializer())
        np.testing.assert_allclose(s.run(v.value()),
                                   np.arange(6).reshape(2, 3))
        s.run(v.assign_add(np.ones((2, 3), np.float32)))
        np.testing.assert_allclose(s.run(v.value()),
                                   np.arange(6).reshape(2, 3) + 1)


def test_gradients_simple():
    with tf.Session() as s:
        x = tf.placeholder(tf.float32, [3])
        loss = tf.reduce_sum(tf.square(x))
        (g,) = tf.gradients(loss, [x])
        out = s.run(g, feed_dict={x: np.array([1, -2, 3], np.float32)})
    np.testing.assert_allclose(out, [2, -4, 6])


def test_gradients_matmul_chain():
    a = np.random.randn(4, 3).astype(np.float32)
    w_np = np.random.randn(3, 2).astype(np.float32)
    with tf.Session() as s:
        x = tf.constant(a)
        w = tf.Variable(w_np)
        y = tf.matmul(x, w)
        loss = tf.reduce_sum(y * y)
        (gw,) = tf.gradients(loss, [w])
        s.run(tf.global_variables_initializer())
        out = s.run(gw)
    expected = 2 * a.T @ (a @ w_np)
    np.testing.assert_allclose(out, expected, rtol=1e-4)


def test_numeric_gradient_check():
    """Numeric vs symbolic jacobian (gradient_checker analog)."""
    np.random.seed(0)
    x0 = np.random.randn(5).astype(np.float32)
    with tf.Session() as s:
        x = tf.placeholder(tf.float32, [5])
        y = tf.reduce_sum(tf.sigmoid(x) * tf.tanh(x))
        (g,) = tf.gradients(y, [x])
        sym = s.run(g, feed_dict={x: x0})
        eps = 1e-3
        num = np.zeros(5, np.float32)
        for i in range(5):
            xp = x0.copy(); xp[i] += eps
            xm = x0.copy(); xm[i] -= eps
            num[i] = (s.run(y, feed_dict={x: xp}) -
                      s.run(y, feed_dict={x: xm})) / (2 * eps)
    np.testing.assert_allclose(sym, num, rtol=1e-2, atol=1e-3)


def test_cond():
    with tf.Session() as s:
        p = tf.placeholder(tf.bool, [])
        c = tf.cond(p, lambda: tf.constant(1.0) * 3.0,
                    lambda: tf.constant(2.0) + 4.0)
        assert s.run(c, feed_dict={p: np.array(True)}) == 3.0
        assert s.run(c, feed_dict={p: np.array(False)}) == 6.0


def test_while_loop():
    with tf.Session() as s:
        i = tf.constant(0)
        r = tf.while_loop(lambda i: tf.less(i, 10),
                          lambda i: tf.add(i, 1), [i])
        assert s.run(r) == 10


def test_while_loop_multivar():
    with tf.Session() as s:
        i = tf.constant(0)
        acc = tf.constant(1.0)
        r = tf.while_loop(lambda i, a: tf.less(i, 8),
                          lambda i, a: (tf.add(i, 1), a * 2.0), [i, acc])
        i_out, a_out = s.run(r)
        assert i_out == 8
        assert a_out == 256.0


def test_zero_iteration_loop():
    with tf.Session() as s:
        i = tf.constant(42)
        r = tf.while_loop(lambda i: tf.less(i, 0),
                          lambda i: tf.add(i, 1), [i])
        assert s.run(r) == 42


def test_softmax_xent_vs_numpy():
    np.random.seed(1)
    logits = np.random.randn(8, 10).astype(np.float32)
    labels = np.random.randint(0, 10, 8)
    onehot = np.eye(10, dtype=np.float32)[labels]
    with tf.Session() as s:
        loss = tf.nn.softmax_cross_entropy_with_logits(
            labels=tf.constant(onehot), logits=tf.constant(logits))
        out = s.run(loss)
    e = np.exp(logits - logits.max(1, keepdims=True))
    p = e / e.sum(1, keepdims=True)
    expected = -(onehot * np.log(p)).sum(1)
    np.testing.assert_allclose(out, expected, rtol=1e-5)


def test_conv2d_vs_numpy():
    np.random.seed(2)
    x = np.random.randn(2, 8, 8, 3).astype(np.float32)
    w = np.random.randn(3, 3, 3, 4).astype(np.float32)
    with tf.Session() as s:
        y = tf.nn.conv2d(tf.constant(x), tf.constant(w),
                         strides=[1, 1, 1, 1], padding='VALID')
        out = s.run(y)
    ref = np.zeros((2, 6, 6, 4), np.float32)
    for n in range(2):
        for i in range(6):
            for j in range(6):
                patch = x[n, i:i+3, j:j+3, :]
                ref[n, i, j] = np.tensordot(patch, w, axes=3)
    np.testing.assert_allclose(out, ref, rtol=1e-4, atol=1e-4)


def test_optimizer_sgd_quadratic():
    with tf.Session() as s:
        v = tf.Variable(5.0)
        loss = tf.square(v)
        opt = tf.train.GradientDescentOptimizer(0.1)
        train = opt.minimize(loss)
        s.run(tf.global_variables_initializer())
        for _ in range(50):
            s.run(train)
        assert np.abs(s.run(v.value())) < 0.1


def test_string_constant():
    with tf.Session() as s:
        c = tf.constant('hello')
        assert s.run(c) == b'hello'
