"""CTC loss / greedy decoder (csrc/kernels/cpu_ctc.cc; reference
core/kernels/ctc_loss_op.cc analog). Loss is checked against an independent
numpy forward-pass; the gradient against numeric differentiation."""
import numpy as np

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.ops import nn_ops, sparse_ops


def setup_function(_):
    tf.reset_default_graph()
    np.random.seed(11)


def _np_ctc_loss(logits_tb, labels, blank):
    """Reference forward pass for ONE batch item: logits [T, C]."""
    T, C = logits_tb.shape
    p = np.exp(logits_tb - logits_tb.max(1, keepdims=True))
    p /= p.sum(1, keepdims=True)
    ext = [blank]
    for l in labels:
        ext += [l, blank]
    S = len(ext)
    alpha = np.zeros((T, S))
    alpha[0, 0] = p[0, ext[0]]
    if S > 1:
        alpha[0, 1] = p[0, ext[1]]
    for t in range(1, T):
        for s in range(S):
            a = alpha[t - 1, s]
            if s > 0:
                a += alpha[t - 1, s - 1]
            if s > 1 and ext[s] != blank and ext[s] != ext[s - 2]:
                a += alpha[t - 1, s - 2]
            alpha[t, s] = a * p[t, ext[s]]
    ll = alpha[T - 1, S - 1] + (alpha[T - 1, S - 2] if S > 1 else 0.0)
    return -np.log(ll)


def test_ctc_loss_matches_numpy():
    T, B, C = 8, 3, 6
    logits = np.random.randn(T, B, C).astype(np.float32)
    label_list = [[1, 2, 3], [0, 0], [4]]
    idx, vals = [], []
    for b, ls in enumerate(label_list):
        for t, l in enumerate(ls):
            idx.append([b, t])
            vals.append(l)
    labels = sparse_ops.SparseTensor(
        np.array(idx, np.int64), np.array(vals, np.int32),
        np.array([B, 3], np.int64))
    loss = nn_ops.ctc_loss(labels, tf.constant(logits), [T] * B)
    with tf.Session() as s:
        got = s.run(loss)
    for b in range(B):
        want = _np_ctc_loss(logits[:, b, :], label_list[b], C - 1)
        np.testing.assert_allclose(got[b], want, rtol=1e-4)


def test_ctc_loss_gradient_numeric():
    T, B, C = 5, 1, 4
    logits = np.random.randn(T, B, C).astype(np.float32)
    labels = sparse_ops.SparseTensor(
        np.array([[0, 0], [0, 1]], np.int64), np.array([0, 2], np.int32),
        np.array([1, 2], np.int64))
    ph = tf.placeholder(tf.float32, [T, B, C])
    loss = nn_ops.ctc_loss(labels, ph, [T])
    g = tf.gradients(tf.reduce_sum(loss), [ph])[0]
    with tf.Session() as s:
        gv = s.run(g, {ph: logits})
        eps = 1e-3
        for t in range(T):
            for c in range(C):
                xp = logits.copy()
                xp[t, 0, c] += eps
                xm = logits.copy()
                xm[t, 0, c] -= eps
                num = (s.run(loss, {ph: xp})[0] -
                       s.run(loss, {ph: xm})[0]) / (2 * eps)
                np.testing.assert_allclose(gv[t, 0, c], num, rtol=2e-2,
                                           atol=2e-3)


def test_ctc_greedy_decoder():
    T, B, C = 6, 2, 4
    logits = np.full((T, B, C), -5.0, np.float32)
    # batch 0 greedy path: 1 1 blank 2 2 0 -> merged decode [1, 2, 0]
    path0 = [1, 1, 3, 2, 2, 0]
    for t, c in enumerate(path0):
        logits[t, 0, c] = 5.0
    # batch 1: all blanks -> empty
    for t in range(T):
        logits[t, 1, 3] = 5.0
    decoded, logp = nn_ops.ctc_greedy_decoder(tf.constant(logits), [T, T])
    st = decoded[0]
    with tf.Session() as s:
        idx, vals, shape = s.run([st.indices, st.values, st.dense_shape])
    assert vals.tolist() == [1, 2, 0]
    assert idx.tolist() == [[0, 0], [0, 1], [0, 2]]
    assert shape.tolist() == [2, 3]


def test_ctc_label_too_long_errors():
    import pytest
    T, B, C = 2, 1, 4
    logits = np.random.randn(T, B, C).astype(np.float32)
    labels = sparse_ops.SparseTensor(
        np.array([[0, 0], [0, 1], [0, 2]], np.int64),
        np.array([0, 1, 2], np.int32), np.array([1, 3], np.int64))
    loss = nn_ops.ctc_loss(labels, tf.constant(logits), [T])
    with tf.Session() as s:
        with pytest.raises(Exception, match='not enough time steps'):
            s.run(loss)
