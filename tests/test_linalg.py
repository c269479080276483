"""Dense linalg ops vs numpy (csrc/kernels/cpu_linalg.cc; reference
core/ops/linalg_ops.cc + python/ops/linalg_grad.py analogs)."""
import numpy as np
import pytest

import simple_tensorflow_amd as tf


def setup_function(_):
    tf.reset_default_graph()
    np.random.seed(7)


def _spd(n):
    a = np.random.randn(n, n)
    return a @ a.T + n * np.eye(n)


def _run(t):
    with tf.Session() as s:
        return s.run(t)


def test_cholesky():
    A = _spd(6)
    L = _run(tf.cholesky(tf.constant(A, dtype=tf.float64)))
    np.testing.assert_allclose(L @ L.T, A, atol=1e-10)
    assert np.allclose(np.triu(L, 1), 0)


def test_cholesky_batched():
    A = np.stack([_spd(4), _spd(4)])
    L = _run(tf.cholesky(tf.constant(A, dtype=tf.float64)))
    for b in range(2):
        np.testing.assert_allclose(L[b] @ L[b].T, A[b], atol=1e-10)


def test_cholesky_not_spd_errors():
    A = np.eye(3)
    A[2, 2] = -1.0
    with pytest.raises(Exception, match='positive definite'):
        _run(tf.cholesky(tf.constant(A, dtype=tf.float64)))


def test_determinant_and_inverse():
    A = _spd(5)
    cA = tf.constant(A, dtype=tf.float64)
    d, inv = _run([tf.matrix_determinant(cA), tf.matrix_inverse(cA)])
    np.testing.assert_allclose(d, np.linalg.det(A), rtol=1e-10)
    np.testing.assert_allclose(inv, np.linalg.inv(A), atol=1e-10)
    invT = _run(tf.matrix_inverse(cA + tf.constant(
        np.triu(np.ones((5, 5)), 1), dtype=tf.float64), adjoint=True))
    B = A + np.triu(np.ones((5, 5)), 1)
    np.testing.assert_allclose(invT, np.linalg.inv(B.T), atol=1e-9)


def test_matrix_solve():
    A = _spd(5)
    B = np.random.randn(5, 3)
    x = _run(tf.matrix_solve(tf.constant(A, dtype=tf.float64),
                             tf.constant(B, dtype=tf.float64)))
    np.testing.assert_allclose(x, np.linalg.solve(A, B), atol=1e-10)
    xa = _run(tf.matrix_solve(tf.constant(A + np.triu(np.ones((5, 5)), 1),
                                          dtype=tf.float64),
                              tf.constant(B, dtype=tf.float64), adjoint=True))
    np.testing.assert_allclose(
        xa, np.linalg.solve((A + np.triu(np.ones((5, 5)), 1)).T, B),
        atol=1e-9)


def test_triangular_solve():
    L = np.tril(np.random.randn(5, 5)) + 5 * np.eye(5)
    B = np.random.randn(5, 2)
    x = _run(tf.matrix_triangular_solve(tf.constant(L, dtype=tf.float64),
                                        tf.constant(B, dtype=tf.float64)))
    np.testing.assert_allclose(L @ x, B, atol=1e-10)
    U = L.T
    xu = _run(tf.matrix_triangular_solve(tf.constant(U, dtype=tf.float64),
                                         tf.constant(B, dtype=tf.float64),
                                         lower=False))
    np.testing.assert_allclose(U @ xu, B, atol=1e-10)


def test_qr():
    M = np.random.randn(6, 4)
    q, r = _run(list(tf.qr(tf.constant(M, dtype=tf.float64))))
    assert q.shape == (6, 4) and r.shape == (4, 4)
    np.testing.assert_allclose(q @ r, M, atol=1e-10)
    np.testing.assert_allclose(q.T @ q, np.eye(4), atol=1e-10)
    assert np.allclose(np.tril(r, -1), 0)
    qf, rf = _run(list(tf.qr(tf.constant(M, dtype=tf.float64),
                             full_matrices=True)))
    assert qf.shape == (6, 6) and rf.shape == (6, 4)
    np.testing.assert_allclose(qf @ rf, M, atol=1e-10)


def test_svd():
    for shape in [(6, 4), (4, 6), (5, 5)]:
        M = np.random.randn(*shape)
        s, u, v = _run(list(tf.svd(tf.constant(M, dtype=tf.float64))))
        np.testing.assert_allclose(s, np.linalg.svd(M, compute_uv=False),
                                   atol=1e-9)
        np.testing.assert_allclose(u @ np.diag(s) @ v.T, M, atol=1e-8)
    s_only = _run(tf.svd(tf.constant(np.random.randn(4, 3),
                                     dtype=tf.float64), compute_uv=False))
    assert s_only.shape == (3,)


def test_self_adjoint_eig():
    A = _spd(6)
    e, v = _run(list(tf.self_adjoint_eig(tf.constant(A, dtype=tf.float64))))
    np.testing.assert_allclose(e, np.linalg.eigvalsh(A), atol=1e-9)
    np.testing.assert_allclose(A @ v, v @ np.diag(e), atol=1e-8)


def test_matrix_solve_ls():
    M = np.random.randn(8, 4)
    B = np.random.randn(8, 2)
    x = _run(tf.matrix_solve_ls(tf.constant(M, dtype=tf.float64),
                                tf.constant(B, dtype=tf.float64)))
    want, *_ = np.linalg.lstsq(M, B, rcond=None)
    np.testing.assert_allclose(x, want, atol=1e-8)


def _numeric_grad(f, x, eps=1e-6):
    g = np.zeros_like(x)
    it = np.nditer(x, flags=['multi_index'])
    while not it.finished:
        idx = it.multi_index
        xp = x.copy()
        xp[idx] += eps
        xm = x.copy()
        xm[idx] -= eps
        g[idx] = (f(xp) - f(xm)) / (2 * eps)
        it.iternext()
    return g


def test_matrix_inverse_grad():
    A = _spd(4)
    ph = tf.placeholder(tf.float64, [4, 4])
    loss = tf.reduce_sum(tf.matrix_inverse(ph) ** 2.0)
    g = tf.gradients(loss, [ph])[0]
    with tf.Session() as s:
        got = s.run(g, {ph: A})
        num = _numeric_grad(lambda x: s.run(loss, {ph: x}), A)
    np.testing.assert_allclose(got, num, rtol=1e-4, atol=1e-6)


def test_matrix_determinant_grad():
    A = _spd(4)
    ph = tf.placeholder(tf.float64, [4, 4])
    loss = tf.matrix_determinant(ph)
    g = tf.gradients(loss, [ph])[0]
    with tf.Session() as s:
        got = s.run(g, {ph: A})
        num = _numeric_grad(lambda x: s.run(loss, {ph: x}), A)
    np.testing.assert_allclose(got, num, rtol=1e-4, atol=1e-5)


def test_matrix_solve_grad():
    A = _spd(3)
    B = np.random.randn(3, 2)
    pa = tf.placeholder(tf.float64, [3, 3])
    pb = tf.placeholder(tf.float64, [3, 2])
    loss = tf.reduce_sum(tf.matrix_solve(pa, pb) ** 2.0)
    ga, gb = tf.gradients(loss, [pa, pb])
    with tf.Session() as s:
        gota, gotb = s.run([ga, gb], {pa: A, pb: B})
        numa = _numeric_grad(lambda x: s.run(loss, {pa: x, pb: B}), A)
        numb = _numeric_grad(lambda x: s.run(loss, {pa: A, pb: x}), B)
    np.testing.assert_allclose(gota, numa, rtol=1e-4, atol=1e-6)
    np.testing.assert_allclose(gotb, numb, rtol=1e-4, atol=1e-6)


def test_cholesky_gradient():
    A = _spd(4)
    ph = tf.placeholder(tf.float64, [4, 4])
    # symmetrize input so numeric grad matches the symmetric-projection
    # convention of CholeskyGrad
    sym = (ph + tf.transpose(ph)) * 0.5
    loss = tf.reduce_sum(tf.cholesky(sym) ** 2.0)
    g = tf.gradients(loss, [ph])[0]
    with tf.Session() as s:
        got = s.run(g, {ph: A})
        num = _numeric_grad(lambda x: s.run(loss, {ph: x}), A)
    np.testing.assert_allclose(got, num, rtol=1e-4, atol=1e-6)


def test_eye():
    with tf.Session() as s:
        np.testing.assert_allclose(s.run(tf.eye(3)), np.eye(3))
        b = s.run(tf.eye(2, batch_shape=[4]))
        assert b.shape == (4, 2, 2)
