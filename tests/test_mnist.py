"""Milestone A (BASELINE.json config 1): MNIST softmax regression via
tf.Session on the CPU DirectSession — synthetic data (no network access)."""
import numpy as np
import pytest

import simple_tensorflow_amd as tf


@pytest.fixture(autouse=True)
def fresh_graph():
    tf.reset_default_graph()
    yield


def _synthetic_mnist(n, seed=0):
    rng = np.random.RandomState(seed)
    # class-dependent means so the problem is learnable
    labels = rng.randint(0, 10, n)
    x = rng.randn(n, 784).astype(np.float32) * 0.5
    for c in range(10):
        x[labels == c, c * 78:(c + 1) * 78] += 1.5
    return x, labels


def test_mnist_softmax_training_decreases_loss():
    x_np, y_np = _synthetic_mnist(512)
    onehot = np.eye(10, dtype=np.float32)[y_np]

    x = tf.placeholder(tf.float32, [None, 784])
    y_ = tf.placeholder(tf.float32, [None, 10])
    w = tf.Variable(tf.zeros([784, 10]))
    b = tf.Variable(tf.zeros([10]))
    logits = tf.matmul(x, w) + b
    loss = tf.reduce_mean(
        tf.nn.softmax_cross_entropy_with_logits(labels=y_, logits=logits))
    train_step = tf.train.GradientDescentOptimizer(0.5).minimize(loss)
    correct = tf.equal(tf.argmax(logits, 1), tf.argmax(y_, 1))
    accuracy = tf.reduce_mean(tf.cast(correct, tf.float32))

    with tf.Session() as sess:
        sess.run(tf.global_variables_initializer())
        feed = {x: x_np, y_: onehot}
        loss0 = sess.run(loss, feed_dict=feed)
        for _ in range(30):
            sess.run(train_step, feed_dict=feed)
        loss1, acc = sess.run([loss, accuracy], feed_dict=feed)

    assert loss0 == pytest.approx(np.log(10), rel=1e-3)
    assert loss1 < 0.5 * loss0
    assert acc > 0.8


def test_mnist_sparse_xent_with_momentum():
    x_np, y_np = _synthetic_mnist(256, seed=3)
    x = tf.placeholder(tf.float32, [None, 784])
    y_ = tf.placeholder(tf.int64, [None])
    w = tf.Variable(tf.truncated_normal([784, 10], stddev=0.01, seed=7))
    b = tf.Variable(tf.zeros([10]))
    logits = tf.nn.xw_plus_b(x, w, b)
    loss = tf.reduce_mean(tf.nn.sparse_softmax_cross_entropy_with_logits(
        labels=y_, logits=logits))
    train = tf.train.MomentumOptimizer(0.1, 0.9).minimize(loss)
    with tf.Session() as sess:
        sess.run(tf.global_variables_initializer())
        feed = {x: x_np, y_: y_np}
        l0 = sess.run(loss, feed_dict=feed)
        for _ in range(20):
            sess.run(train, feed_dict=feed)
        l1 = sess.run(loss, feed_dict=feed)
    assert l1 < l0
