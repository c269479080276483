"""Dynamic-shape gradients: Concat/Slice/Pad/Tile/Gather grads
built from runtime shape ops (ShapeN/ConcatOffset/Shape), so models with an
unknown batch dimension train end-to-end — matching the reference's
array_grad.py forms."""
import numpy as np
import pytest

import simple_tensorflow_amd as tf


@pytest.fixture(autouse=True)
def fresh_graph():
    tf.reset_default_graph()
    yield


def _feed(shape):
    rng = np.random.RandomState(hash(shape) % 2**31)
    return rng.randn(*shape).astype(np.float32)


def test_concat_grad_dynamic():
    a = tf.placeholder(tf.float32, [None, 3])
    b = tf.placeholder(tf.float32, [None, 3])
    y = tf.concat([a, b], 0)
    w = tf.constant(np.arange(6, dtype=np.float32).reshape(1, 6))
    loss = tf.reduce_sum(y * tf.reshape(w, [2, 3])[0])  # uses slice too
    ga, gb = tf.gradients(loss, [a, b])
    av, bv = _feed((2, 3)), _feed((4, 3))
    with tf.Session() as s:
        ra, rb = s.run([ga, gb], feed_dict={a: av, b: bv})
    assert ra.shape == (2, 3) and rb.shape == (4, 3)
    np.testing.assert_allclose(ra, np.tile([[0, 1, 2]], (2, 1)))
    np.testing.assert_allclose(rb, np.tile([[0, 1, 2]], (4, 1)))


def test_concat_grad_axis1_dynamic():
    a = tf.placeholder(tf.float32, [None, None])
    b = tf.placeholder(tf.float32, [None, None])
    y = tf.concat([a, b], 1)
    loss = tf.reduce_sum(y * y)
    ga, gb = tf.gradients(loss, [a, b])
    av, bv = _feed((3, 2)), _feed((3, 5))
    with tf.Session() as s:
        ra, rb = s.run([ga, gb], feed_dict={a: av, b: bv})
    np.testing.assert_allclose(ra, 2 * av, rtol=1e-6)
    np.testing.assert_allclose(rb, 2 * bv, rtol=1e-6)


def test_slice_grad_dynamic():
    x = tf.placeholder(tf.float32, [None, 4])
    y = tf.slice(x, [1, 1], [-1, 2])
    loss = tf.reduce_sum(y)
    g = tf.gradients(loss, [x])[0]
    xv = _feed((3, 4))
    with tf.Session() as s:
        r = s.run(g, feed_dict={x: xv})
    want = np.zeros((3, 4), np.float32)
    want[1:, 1:3] = 1.0
    np.testing.assert_allclose(r, want)


def test_pad_grad_dynamic():
    x = tf.placeholder(tf.float32, [None, 2])
    y = tf.pad(x, [[1, 2], [0, 1]])
    loss = tf.reduce_sum(y * y)
    g = tf.gradients(loss, [x])[0]
    xv = _feed((3, 2))
    with tf.Session() as s:
        r = s.run(g, feed_dict={x: xv})
    np.testing.assert_allclose(r, 2 * xv, rtol=1e-6)


def test_tile_grad_dynamic():
    x = tf.placeholder(tf.float32, [None, 2])
    y = tf.tile(x, [3, 2])
    loss = tf.reduce_sum(y)
    g = tf.gradients(loss, [x])[0]
    xv = _feed((2, 2))
    with tf.Session() as s:
        r = s.run(g, feed_dict={x: xv})
    np.testing.assert_allclose(r, np.full((2, 2), 6.0))


def test_gather_grad_dynamic():
    p = tf.placeholder(tf.float32, [None, None])
    idx = tf.constant(np.array([0, 2, 0], np.int32))
    y = tf.gather(p, idx)
    loss = tf.reduce_sum(y)
    g = tf.gradients(loss, [p])[0]
    pv = _feed((4, 3))
    with tf.Session() as s:
        r = s.run(g, feed_dict={p: pv})
    want = np.zeros((4, 3), np.float32)
    want[0] = 2.0
    want[2] = 1.0
    np.testing.assert_allclose(r, want)


def test_transpose_grad_dynamic_perm():
    x = tf.placeholder(tf.float32, [None, None])
    perm = tf.placeholder(tf.int32, [2])
    y = tf.transpose(x, perm)
    loss = tf.reduce_sum(y * tf.constant(np.arange(6, np.float32)
                                         .reshape(3, 2) if False else
                                         np.arange(6).reshape(3, 2)
                                         .astype(np.float32)))
    g = tf.gradients(loss, [x])[0]
    xv = _feed((2, 3))
    with tf.Session() as s:
        r = s.run(g, feed_dict={x: xv, perm: np.array([1, 0], np.int32)})
    want = np.arange(6).reshape(3, 2).T.astype(np.float32)
    np.testing.assert_allclose(r, want)


def test_unknown_batch_model_trains():
    """End-to-end: an MLP with unknown batch dim trains (the VERDICT #4
    'done' criterion)."""
    x = tf.placeholder(tf.float32, [None, 8])
    labels = tf.placeholder(tf.int64, [None])
    w1 = tf.Variable(np.random.RandomState(0)
                     .randn(8, 16).astype(np.float32) * 0.3)
    b1 = tf.Variable(np.zeros(16, np.float32))
    h = tf.nn.relu(tf.matmul(x, w1._as_graph_element()) +
                   b1._as_graph_element())
    # concat a sliced copy to exercise dynamic concat/slice grads in-model
    h2 = tf.concat([h, tf.slice(h, [0, 0], [-1, 8])], 1)
    w2 = tf.Variable(np.random.RandomState(1)
                     .randn(24, 4).astype(np.float32) * 0.3)
    logits = tf.matmul(h2, w2._as_graph_element())
    loss = tf.reduce_mean(tf.nn.sparse_softmax_cross_entropy_with_logits(
        labels=labels, logits=logits))
    train = tf.train.GradientDescentOptimizer(0.5).minimize(loss)
    rng = np.random.RandomState(7)
    teacher = rng.randn(8, 4).astype(np.float32)  # learnable linear task
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        losses = []
        for i in range(60):
            bs = 16 if i % 2 == 0 else 24  # batch size varies run to run
            xv = rng.randn(bs, 8).astype(np.float32)
            yv = np.argmax(xv @ teacher, 1).astype(np.int64)
            _, lv = s.run([train, loss], feed_dict={x: xv, labels: yv})
            losses.append(lv)
    assert np.mean(losses[-5:]) < np.mean(losses[:5]) * 0.7, losses
