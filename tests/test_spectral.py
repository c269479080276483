"""FFT family vs numpy.fft (csrc/kernels/cpu_fft.cc; reference
core/ops/spectral_ops.cc analog — radix-2 + Bluestein)."""
import numpy as np

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.ops import spectral_ops


def setup_function(_):
    tf.reset_default_graph()
    np.random.seed(3)


def _cx(*shape):
    return (np.random.randn(*shape) +
            1j * np.random.randn(*shape)).astype(np.complex64)


def _run(t):
    with tf.Session() as s:
        return s.run(t)


def test_fft_pow2_and_roundtrip():
    x = _cx(3, 16)
    y = _run(tf.fft(tf.constant(x)))
    np.testing.assert_allclose(y, np.fft.fft(x), rtol=1e-4, atol=1e-4)
    back = _run(tf.ifft(tf.constant(y)))
    np.testing.assert_allclose(back, x, rtol=1e-4, atol=1e-5)


def test_fft_non_pow2_bluestein():
    for n in (7, 12, 15, 100):
        x = _cx(2, n)
        y = _run(tf.fft(tf.constant(x)))
        np.testing.assert_allclose(y, np.fft.fft(x), rtol=1e-3, atol=1e-3)


def test_fft2d_3d():
    x = _cx(4, 6, 10)
    np.testing.assert_allclose(_run(tf.fft2d(tf.constant(x))),
                               np.fft.fft2(x), rtol=1e-3, atol=1e-3)
    np.testing.assert_allclose(_run(tf.ifft2d(tf.constant(x))),
                               np.fft.ifft2(x), rtol=1e-3, atol=1e-5)
    x3 = _cx(2, 4, 6, 8)
    np.testing.assert_allclose(_run(tf.fft3d(tf.constant(x3))),
                               np.fft.fftn(x3, axes=(-3, -2, -1)),
                               rtol=1e-3, atol=1e-3)


def test_rfft_irfft():
    x = np.random.randn(3, 16).astype(np.float32)
    y = _run(spectral_ops.rfft(x))
    np.testing.assert_allclose(y, np.fft.rfft(x), rtol=1e-4, atol=1e-4)
    back = _run(spectral_ops.irfft(y, [16]))
    np.testing.assert_allclose(back, x, rtol=1e-4, atol=1e-5)
    # padded length
    yp = _run(spectral_ops.rfft(x, [32]))
    np.testing.assert_allclose(yp, np.fft.rfft(x, 32), rtol=1e-4, atol=1e-4)


def test_rfft2d_irfft2d():
    x = np.random.randn(2, 8, 12).astype(np.float32)
    y = _run(spectral_ops.rfft2d(x))
    np.testing.assert_allclose(y, np.fft.rfft2(x), rtol=1e-3, atol=1e-3)
    back = _run(spectral_ops.irfft2d(y, [8, 12]))
    np.testing.assert_allclose(back, x, rtol=1e-3, atol=1e-4)


def test_complex_accessors():
    r = np.random.randn(5).astype(np.float32)
    i = np.random.randn(5).astype(np.float32)
    c = tf.complex(tf.constant(r), tf.constant(i))
    with tf.Session() as s:
        cv, rv, iv, conj_v, av = s.run(
            [c, tf.real(c), tf.imag(c), tf.conj(c),
             spectral_ops.complex_abs(c)])
    np.testing.assert_allclose(cv, r + 1j * i, rtol=1e-6)
    np.testing.assert_allclose(rv, r, rtol=1e-6)
    np.testing.assert_allclose(iv, i, rtol=1e-6)
    np.testing.assert_allclose(conj_v, r - 1j * i, rtol=1e-6)
    np.testing.assert_allclose(av, np.abs(r + 1j * i), rtol=1e-5)


def test_fft_gradient():
    # loss = sum(real(fft(complex(x, 0)))^2 + imag(...)^2) — check against
    # numeric differentiation
    x0 = np.random.randn(8).astype(np.float32)
    ph = tf.placeholder(tf.float32, [8])
    f = tf.fft(tf.complex(ph, tf.zeros_like(ph)))
    loss = tf.reduce_sum(tf.real(f) ** 2.0 + tf.imag(f) ** 2.0)
    g = tf.gradients(loss, [ph])[0]
    with tf.Session() as s:
        got = s.run(g, {ph: x0})
        eps = 1e-3
        num = np.zeros(8, np.float32)
        for k in range(8):
            xp, xm = x0.copy(), x0.copy()
            xp[k] += eps
            xm[k] -= eps
            num[k] = (s.run(loss, {ph: xp}) - s.run(loss, {ph: xm})) / (2 * eps)
    np.testing.assert_allclose(got, num, rtol=1e-2, atol=0.3)
