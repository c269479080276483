"""tf.einsum vs numpy.einsum (python/ops/special_math_ops.py; reference
special_math_ops analog)."""
import numpy as np

import simple_tensorflow_amd as tf


def setup_function(_):
    tf.reset_default_graph()
    np.random.seed(0)


def _check(eq, *arrays, **kw):
    tensors = [tf.constant(a) for a in arrays]
    with tf.Session() as s:
        got = s.run(tf.einsum(eq, *tensors))
    want = np.einsum(eq, *arrays)
    np.testing.assert_allclose(got, want, rtol=1e-4, atol=1e-5)


def test_matmul_forms():
    a = np.random.randn(4, 5).astype(np.float32)
    b = np.random.randn(5, 6).astype(np.float32)
    _check('ij,jk->ik', a, b)
    _check('ij,kj->ik', a, np.random.randn(6, 5).astype(np.float32))
    _check('ji,jk->ik', np.random.randn(5, 4).astype(np.float32), b)


def test_batch_matmul():
    a = np.random.randn(3, 4, 5).astype(np.float32)
    b = np.random.randn(3, 5, 6).astype(np.float32)
    _check('bij,bjk->bik', a, b)
    _check('bij,bkj->bik', a, np.random.randn(3, 6, 5).astype(np.float32))


def test_outer_and_inner():
    a = np.random.randn(4).astype(np.float32)
    b = np.random.randn(5).astype(np.float32)
    _check('i,j->ij', a, b)
    _check('i,i->', a, np.random.randn(4).astype(np.float32))


def test_transpose_and_reduce():
    a = np.random.randn(3, 4, 5).astype(np.float32)
    _check('ijk->kji', a)
    _check('ijk->ik', a)
    _check('ijk->', a)


def test_attention_shapes():
    q = np.random.randn(2, 8, 16).astype(np.float32)
    k = np.random.randn(2, 10, 16).astype(np.float32)
    _check('bqd,bkd->bqk', q, k)
    attn = np.random.randn(2, 8, 10).astype(np.float32)
    v = np.random.randn(2, 10, 16).astype(np.float32)
    _check('bqk,bkd->bqd', attn, v)


def test_implicit_output():
    a = np.random.randn(4, 5).astype(np.float32)
    b = np.random.randn(5, 6).astype(np.float32)
    _check('ij,jk', a, b)  # implicit -> 'ik'


def test_gradient_through_einsum():
    a0 = np.random.randn(3, 4).astype(np.float32)
    b0 = np.random.randn(4, 2).astype(np.float32)
    pa = tf.placeholder(tf.float32, [3, 4])
    pb = tf.placeholder(tf.float32, [4, 2])
    loss = tf.reduce_sum(tf.einsum('ij,jk->ik', pa, pb) ** 2.0)
    ga, gb = tf.gradients(loss, [pa, pb])
    with tf.Session() as s:
        gav, gbv = s.run([ga, gb], {pa: a0, pb: b0})
    c = a0 @ b0
    np.testing.assert_allclose(gav, 2 * c @ b0.T, rtol=1e-4)
    np.testing.assert_allclose(gbv, 2 * a0.T @ c, rtol=1e-4)
