"""Scoped elementwise fusion (graph/optimizer.cc FuseElementwise +
_FusedElementwise interpreter kernels; the SURVEY §7.10 fusion slot).
Numerics must match the unfused graph exactly and the fused node must be
what actually executes."""
import os
import subprocess
import sys

import numpy as np
import pytest

import simple_tensorflow_amd as tf


def setup_function(_):
    tf.reset_default_graph()


def _trace_run(fetch, feed):
    opts = tf.RunOptions(trace_level=tf.RunOptions.FULL_TRACE)
    md = tf.RunMetadata()
    with tf.Session() as s:
        out = s.run(fetch, feed, options=opts, run_metadata=md)
    names = [ns.node_name for ds in md.step_stats.dev_stats
             for ns in ds.node_stats]
    return out, names


def test_swish_chain_fuses():
    np.random.seed(0)
    x = np.random.randn(256).astype(np.float32)
    ph = tf.placeholder(tf.float32, [256])
    z = ph * tf.sigmoid(ph) * tf.constant(1.5)
    out, names = _trace_run(z, {ph: x})
    want = x * (1 / (1 + np.exp(-x))) * 1.5
    np.testing.assert_allclose(out, want, rtol=1e-5, atol=1e-6)
    assert any('_fused' in n for n in names)
    assert not any('Sigmoid' in n for n in names)


def test_multi_exit_group():
    np.random.seed(1)
    x = np.random.randn(64).astype(np.float32)
    ph = tf.placeholder(tf.float32, [64])
    a = tf.tanh(ph) * ph          # exit 1 (feeds both fetches)
    b = a + tf.nn.relu(ph)        # fetch
    out_a, out_b = None, None
    with tf.Session() as s:
        out_a, out_b = s.run([a, b], {ph: x})
    want_a = np.tanh(x) * x
    np.testing.assert_allclose(out_a, want_a, rtol=1e-5, atol=1e-6)
    np.testing.assert_allclose(out_b, want_a + np.maximum(x, 0), rtol=1e-5,
                               atol=1e-6)


def test_bf16_chain():
    np.random.seed(2)
    x = np.random.randn(128).astype(np.float32)
    ph = tf.placeholder(tf.bfloat16, [128])
    z = tf.exp(-tf.square(ph)) * ph
    out, names = _trace_run(z, {ph: x})
    assert any('_fused' in n for n in names)
    xb = x.astype(np.float32)
    want = np.exp(-(xb ** 2)) * xb
    np.testing.assert_allclose(np.asarray(out, np.float32), want, rtol=0.05,
                               atol=0.02)


def test_fusion_matches_unfused():
    # run the same chain in a subprocess with fusion disabled; outputs of
    # the fused and unfused graphs must agree to float tolerance
    code = r'''
import sys; sys.path.insert(0, %r)
import numpy as np
import simple_tensorflow_amd as tf
np.random.seed(5)
x = np.random.randn(512).astype(np.float32)
ph = tf.placeholder(tf.float32, [512])
z = tf.nn.relu(ph * tf.sigmoid(ph) + tf.constant(0.25)) * tf.tanh(ph)
with tf.Session() as s:
    print(repr(float(np.sum(s.run(z, {ph: x})))))
'''
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    outs = []
    for no_fusion in ('0', '1'):
        env = dict(os.environ)
        if no_fusion == '1':
            env['STF_NO_FUSION'] = '1'
        else:
            env.pop('STF_NO_FUSION', None)
        r = subprocess.run([sys.executable, '-c', code % root], env=env,
                           capture_output=True, text=True, timeout=120)
        assert r.returncode == 0, r.stderr
        outs.append(float(r.stdout.strip()))
    assert abs(outs[0] - outs[1]) < 1e-3 * max(1.0, abs(outs[1]))


def test_broadcast_not_fused():
    # a binary with two distinct non-scalar roots must NOT be fused
    np.random.seed(3)
    a = np.random.randn(4, 8).astype(np.float32)
    b = np.random.randn(8).astype(np.float32)  # broadcasts over rows
    pa = tf.placeholder(tf.float32, [4, 8])
    pb = tf.placeholder(tf.float32, [8])
    z = tf.tanh(pa) + pb * tf.constant(2.0)
    out, names = _trace_run(z, {pa: a, pb: b})
    np.testing.assert_allclose(out, np.tanh(a) + b * 2.0, rtol=1e-5,
                               atol=1e-6)


@pytest.mark.gpu
def test_fused_kernel_on_gpu():
    np.random.seed(7)
    x = np.random.randn(4096).astype(np.float32)
    ph = tf.placeholder(tf.float32, [4096])
    z = tf.nn.relu(ph * tf.sigmoid(ph) + tf.constant(0.1)) * ph
    out, names = _trace_run(z, {ph: x})
    sig = 1 / (1 + np.exp(-x))
    want = np.maximum(x * sig + 0.1, 0) * x
    np.testing.assert_allclose(out, want, rtol=1e-4, atol=1e-5)
    assert any('_fused' in n for n in names)
