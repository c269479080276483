"""Round-2 breadth composites: masks, counting, convolution wrappers,
multinomial, space/batch, Assert/Print, Ftrl/ProximalSGD, decays
(python/ops/more_ops.py + check_ops.py + optimizer additions)."""
import numpy as np
import pytest

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.framework import errors


def setup_function(_):
    tf.reset_default_graph()
    np.random.seed(0)


def _run(t, feed=None):
    with tf.Session() as s:
        return s.run(t, feed)


def test_boolean_and_sequence_mask():
    bm = _run(tf.boolean_mask(tf.constant(np.arange(5, dtype=np.float32)),
                              tf.constant(np.array([1, 0, 1, 0, 1],
                                                   bool))))
    np.testing.assert_allclose(bm, [0, 2, 4])
    sm = _run(tf.sequence_mask(np.array([1, 3], np.int32), 4))
    assert sm.tolist() == [[True, False, False, False],
                           [True, True, True, False]]


def test_bincount_confusion_matrix():
    assert _run(tf.bincount(np.array([1, 1, 3], np.int32))).tolist() == \
        [0, 2, 0, 1]
    cm = _run(tf.confusion_matrix(np.array([0, 1, 1], np.int64),
                                  np.array([0, 1, 0], np.int64),
                                  num_classes=2))
    assert cm.tolist() == [[1, 0], [1, 1]]


def test_conv_wrappers():
    x = np.random.randn(1, 8, 8, 2).astype(np.float32)
    w = np.random.randn(3, 3, 2, 5).astype(np.float32)
    at = _run(tf.nn.atrous_conv2d(tf.constant(x), tf.constant(w), 2,
                                  'SAME'))
    wd = np.zeros((5, 5, 2, 5), np.float32)
    wd[::2, ::2] = w
    ref = _run(tf.nn.conv2d(tf.constant(x), tf.constant(wd), [1, 1, 1, 1],
                            'SAME'))
    np.testing.assert_allclose(at, ref, rtol=1e-4, atol=1e-4)
    c1 = _run(tf.nn.conv1d(tf.constant(np.random.randn(2, 10, 3)
                                       .astype(np.float32)),
                           tf.constant(np.random.randn(3, 3, 4)
                                       .astype(np.float32)), 1, 'SAME'))
    assert c1.shape == (2, 10, 4)
    sep = _run(tf.nn.separable_conv2d(
        tf.constant(x), tf.constant(np.random.randn(3, 3, 2, 2)
                                    .astype(np.float32)),
        tf.constant(np.random.randn(1, 1, 4, 6).astype(np.float32)),
        [1, 1, 1, 1], 'SAME'))
    assert sep.shape == (1, 8, 8, 6)


def test_multinomial_distribution():
    logits = np.log(np.array([[0.8, 0.2]], np.float32))
    mn = _run(tf.multinomial(tf.constant(logits), 2000, seed=3))
    frac = (mn == 0).mean()
    assert 0.7 < frac < 0.9


def test_space_to_batch_roundtrip():
    x = np.random.randn(2, 6, 6, 3).astype(np.float32)
    sb = tf.space_to_batch(tf.constant(x), [[1, 1], [1, 1]], 2)
    bs = tf.batch_to_space(sb, [[1, 1], [1, 1]], 2)
    np.testing.assert_allclose(_run(bs), x)


def test_assert_and_print():
    x = tf.constant(np.array([1., 2.], np.float32))
    with tf.Session() as s:
        s.run(tf.Assert(tf.reduce_all(x > 0), [x]))
        with pytest.raises(errors.InvalidArgumentError,
                           match='assertion failed'):
            s.run(tf.Assert(tf.reduce_all(x > 5), [x]))
        v = s.run(tf.Print(x, [x], message='dbg'))
    np.testing.assert_allclose(v, [1., 2.])


def test_check_numerics_catches_nan():
    bad = tf.constant(np.array([1.0, np.nan], np.float32)) * tf.constant(1.0)
    cn = tf.add_check_numerics_ops()
    with tf.Session() as s:
        with pytest.raises(Exception):
            s.run([bad, cn] if cn is not None else bad)


def test_ftrl_and_proximal_converge():
    X = np.random.randn(64, 4).astype(np.float32)
    Y = X @ np.array([[1.], [2.], [-1.], [0.5]], np.float32)
    for opt in (tf.train.FtrlOptimizer(0.5),
                tf.train.ProximalGradientDescentOptimizer(0.1)):
        tf.reset_default_graph()
        w = tf.Variable(np.zeros((4, 1), np.float32))
        loss = tf.reduce_mean(
            (tf.matmul(tf.constant(X), w.ref()) - tf.constant(Y)) ** 2.0)
        train = opt.minimize(loss)
        with tf.Session() as s:
            s.run(tf.global_variables_initializer())
            l0 = s.run(loss)
            for _ in range(100):
                s.run(train)
            assert s.run(loss) < l0 * 0.3


def test_decays():
    gs = tf.constant(10)
    ned = _run(tf.train.natural_exp_decay(0.1, gs, 10, 0.5))
    itd = _run(tf.train.inverse_time_decay(0.1, gs, 10, 2.0))
    np.testing.assert_allclose(ned, 0.1 * np.exp(-0.5), rtol=1e-5)
    np.testing.assert_allclose(itd, 0.1 / 3.0, rtol=1e-5)


def test_random_shuffle_and_gamma():
    sh = _run(tf.random_shuffle(tf.constant(np.arange(10,
                                                      dtype=np.float32))))
    assert sorted(sh.tolist()) == list(range(10))
    g = _run(tf.random_gamma([2000], 2.0))
    assert 1.6 < g.mean() < 2.4 and (g > 0).all()


def test_weighted_xent_and_crelu():
    x = np.array([-1.0, 0.5], np.float32)
    z = np.array([1.0, 0.0], np.float32)
    got = _run(tf.nn.weighted_cross_entropy_with_logits(
        targets=tf.constant(z), logits=tf.constant(x), pos_weight=2.0))
    l = 1 + (2.0 - 1) * z
    want = (1 - z) * x + l * (np.log1p(np.exp(-np.abs(x))) +
                              np.maximum(-x, 0))
    np.testing.assert_allclose(got, want, rtol=1e-5)
    cr = _run(tf.nn.crelu(tf.constant(np.array([1., -2.], np.float32))))
    np.testing.assert_allclose(cr, [1, 0, 0, 2])
