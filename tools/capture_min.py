"""Bisect the hipGraph-capture corruption: progressively larger no-io steps,
capture-vs-eager final state comparison. Run twice (with and without
STF_NO_HIPGRAPH) and diff the printed values."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.ops import variables

STEPS = 6


def scenario(name, build):
    tf.reset_default_graph()
    op, probe = build()
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        for _ in range(STEPS):
            s.run(op)
        v = s.run(probe)
    print('%-22s %s' % (name, np.array2string(
        np.asarray(v).ravel()[:6], precision=4)), flush=True)


def case_assign_add():
    v = variables.Variable(tf.zeros([4], tf.float32))
    op = tf.assign_add(v.ref(), tf.constant([1., 2., 3., 4.])).op
    return op, v.ref()


def case_matmul_assign():
    rng = np.random.RandomState(5)
    x = tf.constant(rng.randn(4, 4).astype(np.float32))
    v = variables.Variable(tf.zeros([4, 4], tf.float32))
    op = tf.assign_add(v.ref(), tf.matmul(x, x)).op
    return op, v.ref()


def case_sgd_linear():
    rng = np.random.RandomState(5)
    x = tf.constant(rng.randn(16, 8).astype(np.float32))
    y = tf.constant(rng.randn(16, 4).astype(np.float32))
    w = variables.Variable(tf.truncated_normal([8, 4], stddev=0.1, seed=1))
    err = tf.matmul(x, w.ref()) - y
    loss = tf.reduce_mean(err * err)
    op = tf.train.GradientDescentOptimizer(0.01).minimize(loss)
    return op, w.ref()


def case_sgd_xent():
    rng = np.random.RandomState(5)
    x = tf.constant(rng.randn(16, 8).astype(np.float32))
    labels = tf.constant(rng.randint(0, 4, 16).astype(np.int64))
    w = variables.Variable(tf.truncated_normal([8, 4], stddev=0.1, seed=1))
    logits = tf.matmul(x, w.ref())
    loss = tf.reduce_mean(tf.nn.sparse_softmax_cross_entropy_with_logits(
        labels=labels, logits=logits))
    op = tf.train.GradientDescentOptimizer(0.1).minimize(loss)
    return op, w.ref()


def case_sgd_relu_mlp():
    rng = np.random.RandomState(5)
    x = tf.constant(rng.randn(16, 8).astype(np.float32))
    labels = tf.constant(rng.randint(0, 4, 16).astype(np.int64))
    w1 = variables.Variable(tf.truncated_normal([8, 8], stddev=0.1, seed=1))
    w2 = variables.Variable(tf.truncated_normal([8, 4], stddev=0.1, seed=2))
    h = tf.nn.relu(tf.matmul(x, w1.ref()))
    logits = tf.matmul(h, w2.ref())
    loss = tf.reduce_mean(tf.nn.sparse_softmax_cross_entropy_with_logits(
        labels=labels, logits=logits))
    op = tf.train.GradientDescentOptimizer(0.1).minimize(loss)
    return op, w2.ref()


def case_momentum_mlp():
    rng = np.random.RandomState(5)
    x = tf.constant(rng.randn(16, 8).astype(np.float32))
    labels = tf.constant(rng.randint(0, 4, 16).astype(np.int64))
    w1 = variables.Variable(tf.truncated_normal([8, 8], stddev=0.1, seed=1))
    w2 = variables.Variable(tf.truncated_normal([8, 4], stddev=0.1, seed=2))
    h = tf.nn.relu(tf.matmul(x, w1.ref()))
    logits = tf.matmul(h, w2.ref())
    loss = tf.reduce_mean(tf.nn.sparse_softmax_cross_entropy_with_logits(
        labels=labels, logits=logits))
    op = tf.train.MomentumOptimizer(0.05, 0.9).minimize(loss)
    return op, w2.ref()


scenario('assign_add', case_assign_add)
scenario('matmul_assign', case_matmul_assign)
scenario('sgd_linear', case_sgd_linear)
scenario('sgd_xent', case_sgd_xent)
scenario('sgd_relu_mlp', case_sgd_relu_mlp)
scenario('momentum_mlp', case_momentum_mlp)
