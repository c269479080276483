"""losses / metrics / gradient_checker (reference losses_impl.py,
metrics_impl.py, gradient_checker.py analogs)."""
import numpy as np

import simple_tensorflow_amd as tf


def setup_function(_):
    tf.reset_default_graph()


def test_mean_squared_error_loss():
    labels = tf.constant([[1.0], [2.0]])
    preds = tf.constant([[1.5], [1.0]])
    loss = tf.losses.mean_squared_error(labels, preds)
    with tf.Session() as s:
        v = s.run(loss)
    assert abs(v - (0.25 + 1.0) / 2) < 1e-6
    assert len(tf.get_collection(tf.GraphKeys.LOSSES)) == 1


def test_sigmoid_cross_entropy_matches_numpy():
    np.random.seed(0)
    x = np.random.randn(8).astype(np.float32)
    z = (np.random.rand(8) > 0.5).astype(np.float32)
    loss = tf.losses.sigmoid_cross_entropy(tf.constant(z), tf.constant(x))
    with tf.Session() as s:
        v = s.run(loss)
    expect = np.mean(np.maximum(x, 0) - x * z + np.log1p(np.exp(-np.abs(x))))
    assert abs(v - expect) < 1e-5


def test_metrics_accuracy_streaming():
    labels = tf.placeholder(tf.int64, [4])
    preds = tf.placeholder(tf.int64, [4])
    value, update = tf.metrics.accuracy(labels, preds)
    with tf.Session() as s:
        s.run(tf.local_variables_initializer())
        s.run(update, {labels: [1, 2, 3, 4], preds: [1, 2, 0, 0]})  # 2/4
        s.run(update, {labels: [1, 1, 1, 1], preds: [1, 1, 1, 1]})  # 4/4
        v = s.run(value)
    assert abs(v - 6.0 / 8.0) < 1e-6


def test_gradient_checker_matmul():
    x = tf.placeholder(tf.float32, [2, 3])
    w = tf.constant(np.random.RandomState(1).randn(3, 2).astype(np.float32))
    y = tf.matmul(x, w)
    with tf.Session() as s:
        err = tf.test.compute_gradient_error(x, (2, 3), y, (2, 2))
    assert err < 1e-2


def test_gradient_checker_tanh():
    x = tf.placeholder(tf.float32, [5])
    y = tf.tanh(x)
    with tf.Session() as s:
        err = tf.test.compute_gradient_error(x, (5,), y, (5,))
    assert err < 1e-2
