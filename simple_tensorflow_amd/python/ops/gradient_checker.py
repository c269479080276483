"""Numeric-vs-symbolic gradient comparison (reference
python/ops/gradient_checker.py compute_gradient/compute_gradient_error)."""
import numpy as np

from simple_tensorflow_amd.python.framework import ops
from simple_tensorflow_amd.python.ops import gradients_impl


def _numeric_jacobian(x, x_shape, y, y_shape, feed_dict, sess, delta):
    x_size = int(np.prod(x_shape)) if x_shape else 1
    y_size = int(np.prod(y_shape)) if y_shape else 1
    jac = np.zeros((x_size, y_size), dtype=np.float64)
    x_val = feed_dict[x].copy()
    flat = x_val.reshape(-1)
    for i in range(x_size):
        orig = flat[i]
        flat[i] = orig + delta
        feed_dict[x] = x_val
        y_pos = np.asarray(sess.run(y, feed_dict=feed_dict)).reshape(-1)
        flat[i] = orig - delta
        feed_dict[x] = x_val
        y_neg = np.asarray(sess.run(y, feed_dict=feed_dict)).reshape(-1)
        flat[i] = orig
        jac[i, :] = (y_pos - y_neg) / (2.0 * delta)
    feed_dict[x] = x_val
    return jac


def _symbolic_jacobian(x, x_shape, y, y_shape, feed_dict, sess):
    from simple_tensorflow_amd.python.ops import array_ops
    x_size = int(np.prod(x_shape)) if x_shape else 1
    y_size = int(np.prod(y_shape)) if y_shape else 1
    jac = np.zeros((x_size, y_size), dtype=np.float64)
    dy = array_ops.placeholder(y.dtype, list(y_shape))
    gx = gradients_impl.gradients(y, [x], grad_ys=[dy])[0]
    for j in range(y_size):
        seed = np.zeros(y_shape, dtype=np.float32)
        seed.reshape(-1)[j] = 1.0
        fd = dict(feed_dict)
        fd[dy] = seed
        g = np.asarray(sess.run(gx, feed_dict=fd))
        jac[:, j] = g.reshape(-1)
    return jac


def compute_gradient(x, x_shape, y, y_shape, x_init_value=None, delta=1e-3,
                     feed_dict=None, sess=None):
    """Returns (numeric_jacobian, symbolic_jacobian) as [x_size, y_size]."""
    from simple_tensorflow_amd.python.client import session as sess_mod
    s = sess or sess_mod.get_default_session()
    if x_init_value is None:
        x_init_value = np.random.RandomState(0).randn(
            *x_shape).astype(np.float32)
    fd = dict(feed_dict or {})
    fd[x] = x_init_value.astype(np.float32)
    num = _numeric_jacobian(x, x_shape, y, y_shape, fd, s, delta)
    sym = _symbolic_jacobian(x, x_shape, y, y_shape, fd, s)
    return num, sym


def compute_gradient_error(x, x_shape, y, y_shape, x_init_value=None,
                           delta=1e-3, feed_dict=None, sess=None):
    num, sym = compute_gradient(x, x_shape, y, y_shape, x_init_value, delta,
                                feed_dict, sess)
    return float(np.max(np.abs(num - sym)))
