"""Run only the sgd_linear scenario with STF_DEBUG_LAUNCH to trace pointers
across plain runs 1-2, the capture run 3, and probe eager runs."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.ops import variables

rng = np.random.RandomState(5)
x = tf.constant(rng.randn(16, 8).astype(np.float32))
y = tf.constant(rng.randn(16, 4).astype(np.float32))
w = variables.Variable(tf.truncated_normal([8, 4], stddev=0.1, seed=1))
err = tf.matmul(x, w.ref()) - y
loss = tf.reduce_mean(err * err)
op = tf.train.GradientDescentOptimizer(0.01).minimize(loss)
with tf.Session() as s:
    s.run(tf.global_variables_initializer())
    for i in range(5):
        print('=== run', i, file=sys.stderr, flush=True)
        s.run(op)
    print('=== final w fetch', file=sys.stderr, flush=True)
    print('w', s.run(w.ref()).ravel()[:4], flush=True)
