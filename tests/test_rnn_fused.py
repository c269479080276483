"""Fused LSTM cell (LSTMGates / LSTMGatesGrad; csrc HIP LstmGatesKernel,
reference contrib/rnn lstm_ops.cc LSTMBlockCell analog): numerics must match
the composed BasicLSTMCell graph exactly, forward and gradients."""
import numpy as np
import pytest

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.framework.ops import apply_op
from simple_tensorflow_amd.python.ops import init_ops, rnn_cell_impl
from simple_tensorflow_amd.python.ops import variable_scope as vs


def setup_function(_):
    tf.reset_default_graph()


def _sig(x):
    return 1.0 / (1.0 + np.exp(-x))


def test_lstm_gates_forward_vs_numpy():
    np.random.seed(0)
    B, H = 3, 5
    g = np.random.randn(B, 4 * H).astype(np.float32)
    c = np.random.randn(B, H).astype(np.float32)
    outs = apply_op('LSTMGates', tf.constant(g), tf.constant(c),
                    forget_bias=1.0)
    with tf.Session() as s:
        i, f, o, ci, cs, co, h = s.run(list(outs))
    gi, gj, gf, go = np.split(g, 4, axis=1)
    ei, eci, ef, eo = _sig(gi), np.tanh(gj), _sig(gf + 1.0), _sig(go)
    ecs = ef * c + ei * eci
    eco = np.tanh(ecs)
    for got, want in [(i, ei), (f, ef), (o, eo), (ci, eci), (cs, ecs),
                      (co, eco), (h, eo * eco)]:
        np.testing.assert_allclose(got, want, rtol=1e-5, atol=1e-6)


def _build_rnn(cell_cls, W, steps=6, batch=4, in_dim=8, units=16,
               dtype=None):
    tf.reset_default_graph()
    dtype = dtype or tf.float32
    np.random.seed(3)
    x = tf.constant(np.random.randn(batch, steps, in_dim).astype(np.float32),
                    dtype=dtype)
    xs = tf.unstack(x, num=steps, axis=1)
    cell = cell_cls(units)
    state = cell.zero_state(batch, dtype)
    outs = []
    for t, xt in enumerate(xs):
        with vs.variable_scope(
                'rnn', reuse=(t > 0),
                initializer=init_ops.constant_initializer(W) if not t
                else None):
            o, state = cell(xt, state, scope='cell')
        outs.append(o)
    loss = tf.reduce_sum(tf.add_n([tf.reduce_sum(o * o) for o in outs]))
    gvs = tf.train.GradientDescentOptimizer(0.1).compute_gradients(loss)
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        lv = s.run(loss)
        gs = s.run([g for g, _ in gvs if g is not None])
    return lv, gs


def test_fused_cell_matches_composed():
    np.random.seed(1)
    W = (np.random.randn(24, 64) * 0.1).astype(np.float32)
    l1, g1 = _build_rnn(rnn_cell_impl.BasicLSTMCell, W)
    l2, g2 = _build_rnn(rnn_cell_impl.LSTMBlockCell, W)
    assert abs(l1 - l2) < 1e-4 * max(1.0, abs(l1))
    assert len(g1) == len(g2) == 2
    for a, b in zip(g1, g2):
        np.testing.assert_allclose(a, b, rtol=2e-4, atol=2e-5)


def test_fused_cell_bf16_close_to_f32():
    np.random.seed(2)
    W = (np.random.randn(24, 64) * 0.1).astype(np.float32)
    l32, _ = _build_rnn(rnn_cell_impl.LSTMBlockCell, W)
    l16, _ = _build_rnn(rnn_cell_impl.LSTMBlockCell, W, dtype=tf.bfloat16)
    assert abs(float(l16) - l32) < 0.05 * max(1.0, abs(l32))


@pytest.mark.gpu
def test_fused_lstm_gpu_vs_torch():
    torch = pytest.importorskip('torch')
    np.random.seed(4)
    B, H, D, T = 8, 32, 16, 5
    W = (np.random.randn(D + H, 4 * H) * 0.2).astype(np.float32)
    x = np.random.randn(B, T, D).astype(np.float32)

    tf.reset_default_graph()
    xt = tf.constant(x)
    xs = tf.unstack(xt, num=T, axis=1)
    cell = rnn_cell_impl.LSTMBlockCell(H)
    state = cell.zero_state(B, tf.float32)
    outs = []
    for t, xtt in enumerate(xs):
        with vs.variable_scope(
                'rnn', reuse=(t > 0),
                initializer=init_ops.constant_initializer(W) if not t
                else None):
            o, state = cell(xtt, state, scope='cell')
        outs.append(o)
    loss = tf.reduce_sum(tf.add_n([tf.reduce_sum(o * o) for o in outs]))
    gvs = tf.train.GradientDescentOptimizer(0.1).compute_gradients(loss)
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        hs_got = s.run(outs)
        gk_got = s.run([g for g, _ in gvs if g is not None])

    # torch reference: manual LSTM with the same i,j,f,o packed kernel
    Wt = torch.from_numpy(W).requires_grad_(True)
    xtorch = torch.from_numpy(x)
    h = torch.zeros(B, H)
    c = torch.zeros(B, H)
    hs = []
    for t in range(T):
        gates = torch.cat([xtorch[:, t], h], 1) @ Wt
        gi, gj, gf, go = gates.chunk(4, 1)
        i = torch.sigmoid(gi)
        ci = torch.tanh(gj)
        f = torch.sigmoid(gf + 1.0)
        o = torch.sigmoid(go)
        c = f * c + i * ci
        h = o * torch.tanh(c)
        hs.append(h)
    tloss = sum((hh * hh).sum() for hh in hs)
    tloss.backward()
    for got, want in zip(hs_got, hs):
        np.testing.assert_allclose(got, want.detach().numpy(), rtol=1e-4,
                                   atol=1e-5)
    kernel_grad = [g for g in gk_got if g.shape == W.shape][0]
    np.testing.assert_allclose(kernel_grad, Wt.grad.numpy(), rtol=1e-3,
                               atol=1e-4)


@pytest.mark.gpu
def test_bias_grad_skinny_rows_gpu():
    # the LSTM shape that exposed the old one-block BiasGrad: rows=batch=20
    np.random.seed(5)
    dy = np.random.randn(20, 600).astype(np.float32)
    b = tf.Variable(np.zeros(600, np.float32))
    y = tf.nn.bias_add(tf.constant(np.zeros((20, 600), np.float32)), b.ref())
    # d(sum(y * dy))/db = column sums of dy — the actual reduction kernel path
    g = tf.gradients(tf.reduce_sum(y * tf.constant(dy)), [b.ref()])[0]
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        got = s.run(g)
    np.testing.assert_allclose(got, dy.sum(0), rtol=1e-4, atol=1e-3)
