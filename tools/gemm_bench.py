#!/usr/bin/env python3
"""Microbenchmark of the in-tree MFMA GEMM (guide §5.4 rule 24: interleaved
rounds in one process; random uniform [-1,1) data per rule 25)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import simple_tensorflow_amd as tf


def bench_shape(sess_cache, m, n, k, iters=20, dw=False):
    tf.reset_default_graph()
    rng = np.random.RandomState(0)
    if dw:
        # dW shape: C[m,n] = A[k,m]^T x B[k,n] (both contraction-major)
        a = tf.constant((rng.rand(k, m) * 2 - 1).astype(np.float32),
                        dtype=tf.bfloat16)
        b = tf.constant((rng.rand(k, n) * 2 - 1).astype(np.float32),
                        dtype=tf.bfloat16)
        c = tf.matmul(a, b, transpose_a=True)
    else:
        a = tf.constant((rng.rand(m, k) * 2 - 1).astype(np.float32),
                        dtype=tf.bfloat16)
        b = tf.constant((rng.rand(n, k) * 2 - 1).astype(np.float32),
                        dtype=tf.bfloat16)
        c = tf.matmul(a, b, transpose_b=True)  # direct NT
    # keep the result alive on device by snapshotting into a variable
    from simple_tensorflow_amd.python.ops import variables
    v = variables.Variable(tf.zeros([m, n], tf.bfloat16), trainable=False)
    op = tf.assign(v._as_graph_element(), c).op
    s = tf.Session()
    s.run(tf.global_variables_initializer())
    for _ in range(3):
        s.run(op)
    s.sync()
    t0 = time.time()
    for _ in range(iters):
        s.run(op)
    s.sync()
    dt = (time.time() - t0) / iters
    tflops = 2 * m * n * k / dt / 1e12
    print('GEMM %6d x %5d x %5d: %8.3f ms  %8.1f TF/s' %
          (m, n, k, dt * 1e3, tflops), flush=True)
    return tflops


def main():
    import sys
    if len(sys.argv) > 1:
        dw = len(sys.argv) > 2 and sys.argv[2] == 'dw'
        m, n, k = (int(x) for x in sys.argv[1].split(','))
        for _ in range(3):
            bench_shape(None, m, n, k, dw=dw)
        return
    shapes = [
        (4096, 4096, 4096),
        (8192, 8192, 8192),
        (50176, 256, 2304),   # stage-3 3x3 conv fwd
        (802816, 64, 576),    # stage-1 3x3 conv fwd
        (200704, 128, 1152),  # stage-2 3x3
        (12544, 2048, 512),   # stage-4 1x1
        (256, 1000, 2048),    # fc
    ]
    for m, n, k in shapes:
        bench_shape(None, m, n, k)


if __name__ == '__main__':
    main()
