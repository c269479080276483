"""Sweep STF_SPLITK_TARGET for the two hot small-N dW shapes (ResNet-50
b256 stage-1 3x3x64x64 and the 7x7 stem). One process per target value
(PickSplitK caches the env); run via the bash loop in gpurun."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.ops import nn_ops


def bench(shape_name, x_shape, w_shape, strides, iters=30):
    tf.reset_default_graph()
    rng = np.random.RandomState(0)
    x = tf.constant(rng.randn(*x_shape).astype(np.float32),
                    dtype=tf.bfloat16)
    dy_shape = [x_shape[0], x_shape[1] // strides[1],
                x_shape[2] // strides[2], w_shape[3]]
    dy = tf.constant(rng.randn(*dy_shape).astype(np.float32),
                     dtype=tf.bfloat16)
    dw = nn_ops.conv2d_backprop_filter(x, list(w_shape), dy, list(strides),
                                       'SAME')
    out = tf.reduce_sum(tf.cast(dw, tf.float32))
    with tf.Session() as s:
        for _ in range(5):
            s.run(out)
        s.sync()
        t0 = time.perf_counter()
        for _ in range(iters):
            s.run(out)
        s.sync()
        dt = (time.perf_counter() - t0) / iters * 1e3
    print('%s target=%s: %.3f ms' %
          (shape_name, os.environ.get('STF_SPLITK_TARGET', 'default'), dt),
          flush=True)


if __name__ == '__main__':
    which = sys.argv[1] if len(sys.argv) > 1 else 'both'
    if which in ('both', 'stage1'):
        bench('stage1_3x3x64x64', (256, 56, 56, 64), (3, 3, 64, 64),
              (1, 1, 1, 1))
    if which in ('both', 'stem'):
        bench('stem_7x7x3x64', (256, 224, 224, 3), (7, 7, 3, 64),
              (1, 2, 2, 1))
