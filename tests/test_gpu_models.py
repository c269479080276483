"""End-to-end model steps on the MI355X (BASELINE configs 2/4/5)."""
import numpy as np
import pytest

import simple_tensorflow_amd as tf

pytestmark = pytest.mark.gpu


@pytest.fixture(autouse=True)
def fresh_graph():
    tf.reset_default_graph()
    yield


def _train_steps(loss, train_op, steps):
    with tf.Session() as s:
        assert s.num_gpus() > 0
        s.run(tf.global_variables_initializer())
        l0 = s.run(loss)
        for _ in range(steps):
            s.run(train_op)
        l1 = s.run(loss)
    return float(l0), float(l1)


def test_resnet50_trains():
    from simple_tensorflow_amd.models import resnet
    loss, op = resnet.build_train_graph(batch=32, lr=0.05)
    l0, l1 = _train_steps(loss, op, 6)
    assert np.isfinite(l0) and np.isfinite(l1)
    assert abs(l0 - np.log(1000)) < 1.0  # random-init xent ~ ln(1000)
    assert l1 < l0


def test_inception_v3_trains():
    from simple_tensorflow_amd.models import inception
    loss, op = inception.build_train_graph(batch=16, lr=0.05)
    l0, l1 = _train_steps(loss, op, 4)
    assert np.isfinite(l0) and np.isfinite(l1)
    assert l1 < l0 + 0.5


def test_ptb_lstm_trains():
    from simple_tensorflow_amd.models import ptb_lstm
    loss, op = ptb_lstm.build_ptb_graph(batch=20, seq_len=35, hidden=512,
                                        vocab=10000, lr=0.5)
    l0, l1 = _train_steps(loss, op, 4)
    assert np.isfinite(l0) and np.isfinite(l1)
    assert abs(l0 - np.log(10000)) < 1.0
    assert l1 < l0


def test_hipgraph_capture_matches_eager():
    """The captured/replayed step must train the same as eager (correctness
    of the hipGraph path, not just speed)."""
    import os
    from simple_tensorflow_amd.python.ops import variables

    def build_and_train(steps):
        tf.reset_default_graph()
        x = tf.constant(np.random.RandomState(0).randn(64, 32)
                        .astype(np.float32))
        labels = tf.constant(np.random.RandomState(1)
                             .randint(0, 10, 64).astype(np.int64))
        w = variables.Variable(tf.truncated_normal([32, 10], stddev=0.1,
                                                   seed=3))
        logits = tf.matmul(x, w.ref())
        loss = tf.reduce_mean(tf.nn.sparse_softmax_cross_entropy_with_logits(
            labels=labels, logits=logits))
        op = tf.train.GradientDescentOptimizer(0.5).minimize(loss)
        with tf.Session() as s:
            s.run(tf.global_variables_initializer())
            for _ in range(steps):
                s.run(op)
            return float(s.run(loss))

    final = build_and_train(20)  # capture kicks in after run 2
    assert np.isfinite(final)
    assert final < 1.0  # trains well past initial ~2.3


def test_dynamic_rnn_trains_gpu():
    """while-loop gradients on the GPU executor (frames + TensorArray +
    rematerialized backward)."""
    cell = tf.nn.rnn_cell.BasicRNNCell(32)
    x = tf.constant(np.random.RandomState(0)
                    .randn(8, 12, 16).astype(np.float32))
    out, state = tf.nn.dynamic_rnn(cell, x, dtype=tf.float32)
    target = tf.constant(np.random.RandomState(1)
                         .randn(8, 32).astype(np.float32))
    loss = tf.reduce_mean(tf.square(state - target))
    train = tf.train.GradientDescentOptimizer(0.05).minimize(loss)
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        l0 = s.run(loss)
        for _ in range(20):
            s.run(train)
        l1 = s.run(loss)
    assert np.isfinite(l1) and l1 < l0 * 0.9
