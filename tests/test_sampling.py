"""Candidate samplers + sampled losses (csrc/kernels/cpu_sampling.cc;
reference candidate_sampling_ops.cc + nn_impl.py analogs)."""
import numpy as np

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.ops import candidate_sampling_ops as cs


def setup_function(_):
    tf.reset_default_graph()
    np.random.seed(0)


def test_uniform_sampler_unique():
    labels = tf.constant(np.array([[3], [7]], np.int64))
    sampled, true_ec, samp_ec = cs.uniform_candidate_sampler(
        labels, 1, 20, True, 50)
    with tf.Session() as s:
        sv, tec, sec = s.run([sampled, true_ec, samp_ec])
    assert len(set(sv.tolist())) == 20
    assert all(0 <= v < 50 for v in sv)
    assert tec.shape == (2, 1) and sec.shape == (20,)
    assert (tec > 0).all() and (sec > 0).all() and (sec <= 20.5).all()


def test_log_uniform_sampler_skews_low():
    labels = tf.constant(np.array([[0]], np.int64))
    sampled, _, _ = cs.log_uniform_candidate_sampler(
        labels, 1, 2000, False, 10000)
    with tf.Session() as s:
        sv = s.run(sampled)
    # Zipfian: small ids far more frequent than large ids
    low = (sv < 100).sum()
    high = (sv >= 5000).sum()
    assert low > high


def test_accidental_hits():
    labels = tf.constant(np.array([[5], [9]], np.int64))
    sampled = tf.constant(np.array([1, 9, 5, 7], np.int64))
    idx, ids, w = cs.compute_accidental_hits(labels, sampled, 1)
    with tf.Session() as s:
        iv, dv, wv = s.run([idx, ids, w])
    got = set(zip(iv.tolist(), dv.tolist()))
    assert got == {(0, 2), (1, 1)}  # row 0 hits slot 2 (=5), row 1 slot 1
    assert (wv < -1e37).all()


def test_sampled_softmax_trains():
    V, D, B = 60, 8, 4
    w = tf.Variable(np.random.randn(V, D).astype(np.float32) * 0.1)
    b = tf.Variable(np.zeros(V, np.float32))
    x = tf.constant(np.random.randn(B, D).astype(np.float32))
    y = tf.constant(np.random.randint(0, V, (B, 1)).astype(np.int64))
    loss = tf.reduce_mean(
        cs.sampled_softmax_loss(w.ref(), b.ref(), y, x, 15, V))
    opt = tf.train.GradientDescentOptimizer(0.5).minimize(loss)
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        l0 = s.run(loss)
        for _ in range(40):
            s.run(opt)
        l1 = s.run(loss)
    assert l1 < l0 * 0.5


def test_nce_trains():
    V, D, B = 60, 8, 4
    w = tf.Variable(np.random.randn(V, D).astype(np.float32) * 0.1)
    b = tf.Variable(np.zeros(V, np.float32))
    x = tf.constant(np.random.randn(B, D).astype(np.float32))
    y = tf.constant(np.random.randint(0, V, (B, 1)).astype(np.int64))
    loss = tf.reduce_mean(cs.nce_loss(w.ref(), b.ref(), y, x, 15, V))
    opt = tf.train.GradientDescentOptimizer(0.5).minimize(loss)
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        l0 = s.run(loss)
        for _ in range(40):
            s.run(opt)
        l1 = s.run(loss)
    assert l1 < l0
