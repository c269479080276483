"""Quantized ops (csrc/kernels/cpu_quantized.cc; reference quantize_op.cc /
quantized_matmul_op.cc analogs; quint8/qint32 carried as uint8/int32)."""
import numpy as np

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.ops import quantized_ops as q


def setup_function(_):
    tf.reset_default_graph()
    np.random.seed(4)


def test_quantize_dequantize_roundtrip():
    x = np.random.uniform(-3.0, 5.0, 64).astype(np.float32)
    qx, mn, mx = q.quantize_v2(tf.constant(x), -3.0, 5.0)
    back = q.dequantize(qx, mn, mx)
    with tf.Session() as s:
        qv, bv = s.run([qx, back])
    assert qv.dtype == np.uint8
    step = 8.0 / 255
    np.testing.assert_allclose(bv, x, atol=step / 2 + 1e-6)


def test_quantize_clamps():
    x = np.array([-10.0, 0.0, 10.0], np.float32)
    qx, _, _ = q.quantize_v2(tf.constant(x), -1.0, 1.0)
    with tf.Session() as s:
        qv = s.run(qx)
    assert qv[0] == 0 and qv[2] == 255


def test_quantized_matmul_matches_float():
    A = np.random.uniform(-1, 1, (8, 16)).astype(np.float32)
    B = np.random.uniform(-2, 2, (16, 4)).astype(np.float32)
    qa, mna, mxa = q.quantize_v2(tf.constant(A), -1.0, 1.0)
    qb, mnb, mxb = q.quantize_v2(tf.constant(B), -2.0, 2.0)
    qc, mnc, mxc = q.quantized_matmul(qa, qb, mna, mxa, mnb, mxb)
    c = q.dequantize(qc, mnc, mxc)
    with tf.Session() as s:
        got = s.run(c)
    want = A @ B
    # quint8 quantization noise: ~K * step_a * step_b accumulation error
    assert np.abs(got - want).max() < 0.2


def test_quantized_relu():
    x = np.array([-1.0, -0.1, 0.0, 0.5, 1.0], np.float32)
    qx, mn, mx = q.quantize_v2(tf.constant(x), -1.0, 1.0)
    qr, mn2, mx2 = q.quantized_relu(qx, mn, mx)
    back = q.dequantize(qr, mn2, mx2)
    with tf.Session() as s:
        got = s.run(back)
    np.testing.assert_allclose(got, np.maximum(x, 0), atol=2.0 / 255 + 1e-6)


def test_quantize_down_and_requantization_range():
    A = np.random.uniform(-1, 1, (4, 8)).astype(np.float32)
    B = np.random.uniform(-1, 1, (8, 4)).astype(np.float32)
    qa, mna, mxa = q.quantize_v2(tf.constant(A), -1.0, 1.0)
    qb, mnb, mxb = q.quantize_v2(tf.constant(B), -1.0, 1.0)
    qc, mnc, mxc = q.quantized_matmul(qa, qb, mna, mxa, mnb, mxb)
    rmin, rmax = q.requantization_range(qc, mnc, mxc)
    q8, omn, omx = q.quantize_down_and_shrink_range(qc, mnc, mxc)
    deq = q.dequantize(q8, omn, omx)
    with tf.Session() as s:
        got, lo, hi = s.run([deq, rmin, rmax])
    want = A @ B
    assert lo <= want.min() + 0.05 and hi >= want.max() - 0.05
    assert np.abs(got - want).max() < 0.05
