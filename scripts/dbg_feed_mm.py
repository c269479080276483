"""Minimal repro: f32 MatMul with a FED (placeholder) A on GPU."""
import os
import sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np

import simple_tensorflow_amd as tf

rng = np.random.RandomState(0)
x_np = rng.randn(512, 784).astype(np.float32)
w_np = rng.randn(784, 10).astype(np.float32)

xp = tf.placeholder(tf.float32, [None, 784])
W = tf.constant(w_np)
y1 = tf.matmul(xp, W)            # fed A straight into MatMul
y2 = tf.matmul(xp + 0.0, W)      # A passes through a GPU op first
want = x_np @ w_np

with tf.Session() as s:
    v1, v2 = s.run([y1, y2], feed_dict={xp: x_np})
    print('direct-fed diff', np.abs(v1 - want).max())
    print('via-add  diff', np.abs(v2 - want).max())
    bad = np.abs(v1 - want).max(1) > 1e-3
    print('bad rows', bad.sum(), 'first bad', np.argmax(bad) if bad.any()
          else -1)
    if bad.any():
        i = int(np.argmax(bad))
        print('got ', v1[i][:5])
        print('want', want[i][:5])
    # second run same feed
    v1b = s.run(y1, feed_dict={xp: x_np})
    print('run2 direct diff', np.abs(v1b - want).max())
    # small case
    tf.reset_default_graph()
    a_np = rng.randn(8, 16).astype(np.float32)
    b_np = rng.randn(16, 4).astype(np.float32)
    ap = tf.placeholder(tf.float32, [None, 16])
    yy = tf.matmul(ap, tf.constant(b_np))
    with tf.Session() as s2:
        vv = s2.run(yy, feed_dict={ap: a_np})
        print('small diff', np.abs(vv - a_np @ b_np).max())
