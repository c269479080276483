"""Regression tests for round-1 advisor findings.

Each test pins a behavior the reference guarantees:
- tf.gradients always returns a list (reference gradients_impl.py:376).
- All-None grads through a while loop still release upstream producers.
- Max/Min grads split evenly among ties (reference math_grad.py _MaxGrad).
- RestoreV2 verifies crc32c; corrupt checkpoints raise DataLoss.
- Feeds are shape-checked against the placeholder's static shape.
"""
import os

import numpy as np
import pytest

import simple_tensorflow_amd as tf


@pytest.fixture(autouse=True)
def fresh_graph():
    tf.reset_default_graph()
    yield


def test_gradients_single_x_returns_list():
    x = tf.constant([1.0, 2.0, 3.0])
    y = tf.reduce_sum(x * x)
    g = tf.gradients(y, x)
    assert isinstance(g, list) and len(g) == 1
    with tf.Session() as s:
        np.testing.assert_allclose(s.run(g[0]), [2.0, 4.0, 6.0])


def test_gradients_through_stopped_while_loop():
    # y = m*3 + stop_gradient(while_loop body using m): the while pseudo-op
    # receives all-None output grads but must still propagate None so m's
    # other consumer path is differentiated.
    m = tf.constant(2.0)

    def cond(i, acc):
        return tf.less(i, 3)

    def body(i, acc):
        return [tf.add(i, 1), acc * m]

    _, acc = tf.while_loop(cond, body, [tf.constant(0), tf.constant(1.0)])
    y = m * 3.0 + tf.stop_gradient(acc)
    g = tf.gradients(y, [m])
    assert g[0] is not None
    with tf.Session() as s:
        np.testing.assert_allclose(s.run(g[0]), 3.0)


def test_max_grad_splits_ties():
    x = tf.constant([1.0, 1.0, 0.0])
    y = tf.reduce_max(x)
    g = tf.gradients(y, [x])[0]
    with tf.Session() as s:
        np.testing.assert_allclose(s.run(g), [0.5, 0.5, 0.0])


def test_min_grad_splits_ties_2d():
    xv = np.array([[3.0, 1.0, 1.0], [2.0, 2.0, 5.0]], np.float32)
    x = tf.constant(xv)
    y = tf.reduce_min(x, 1)
    g = tf.gradients(tf.reduce_sum(y), [x])[0]
    with tf.Session() as s:
        out = s.run(g)
    np.testing.assert_allclose(out, [[0.0, 0.5, 0.5], [0.5, 0.5, 0.0]])


def test_restore_detects_corruption(tmp_path):
    v = tf.Variable(np.arange(16, dtype=np.float32).reshape(4, 4), name='v')
    saver = tf.train.Saver()
    path = str(tmp_path / 'ckpt')
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        saver.save(s, path)
    shard = None
    for f in os.listdir(str(tmp_path)):
        if '.data-' in f:
            shard = os.path.join(str(tmp_path), f)
    assert shard is not None
    with open(shard, 'r+b') as f:
        f.seek(8)
        b = f.read(1)
        f.seek(8)
        f.write(bytes([b[0] ^ 0xFF]))
    with tf.Session() as s:
        with pytest.raises(tf.errors.DataLossError):
            saver.restore(s, path)


def test_feed_shape_validated():
    p = tf.placeholder(tf.float32, shape=[None, 4])
    y = tf.identity(p)
    with tf.Session() as s:
        out = s.run(y, feed_dict={p: np.zeros((2, 4), np.float32)})
        assert out.shape == (2, 4)
        with pytest.raises(ValueError):
            s.run(y, feed_dict={p: np.zeros((2, 3), np.float32)})
        with pytest.raises(ValueError):
            s.run(y, feed_dict={p: np.zeros((8,), np.float32)})
