"""2-process gloo worker for the data-parallel CPU test: each rank holds the
same variable, computes a rank-dependent gradient, and the DistributedOptimizer
must apply the cross-rank average on both ranks."""
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(
    os.path.dirname(os.path.abspath(__file__)))))

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.parallel import dist
from simple_tensorflow_amd.python.ops import variables


def main():
    world = int(os.environ['WORLD_SIZE'])
    rank = int(os.environ['RANK'])
    # Watchdog: a wedged collective (e.g. two spin-waiting RCCL kernels on
    # one device that never co-schedule) must kill the worker, not the box.
    import signal
    signal.alarm(int(os.environ.get('STF_DIST_WORKER_ALARM', '150')))
    try:
        comm = dist.init(world, rank)
    except Exception as e:  # noqa: BLE001 - report init refusal distinctly
        if 'invalid usage' in str(e).lower() or 'duplicate' in str(e).lower():
            print('RCCL_DUP_UNSUPPORTED:', e)
            return
        raise
    w = variables.Variable(tf.constant([0.0, 0.0]))
    # rank-dependent "loss": grad = [rank+1, 2*(rank+1)]
    coef = tf.constant([float(rank + 1), 2.0 * (rank + 1)])
    loss = tf.reduce_sum(w.ref() * coef)
    opt = dist.DistributedOptimizer(
        tf.train.GradientDescentOptimizer(1.0), world)
    train = opt.minimize(loss)
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        comm.broadcast_variables(s)
        s.run(train)
        v = s.run(w.ref())
    # average grad over ranks: mean(rank+1) (= 1.5 for world=2)
    m = sum(range(1, world + 1)) / world
    expect = -np.array([m, 2.0 * m])
    assert np.allclose(v, expect), (rank, v, expect)
    comm.barrier()

    # ---- multi-bucket: many variables of mixed sizes, tiny bucket cap so
    # several RcclBucketAllReduce ops are built, with deterministic order ----
    tf.reset_default_graph()
    sizes = [3, 1000, 7, 1000, 5, 11]
    vs = [variables.Variable(tf.constant(np.zeros(n, np.float32)))
          for n in sizes]
    loss2 = tf.constant(0.0)
    for i, v_ in enumerate(vs):
        c = tf.constant(np.full(sizes[i], float((rank + 1) * (i + 1)),
                                np.float32))
        loss2 = loss2 + tf.reduce_sum(v_.ref() * c)
    opt2 = dist.DistributedOptimizer(
        tf.train.GradientDescentOptimizer(1.0), world, bucket_bytes=2048)
    train2 = opt2.minimize(loss2)
    nbuckets = len([op for op in tf.get_default_graph().get_operations()
                    if op.type == 'RcclBucketAllReduce'])
    assert nbuckets >= 3, 'expected multiple buckets, got %d' % nbuckets
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        s.run(train2)
        outs = s.run([v_.ref() for v_ in vs])
    mean_rank = (sum(range(1, world + 1)) / world)  # 1.5 for world=2
    for i, out in enumerate(outs):
        expect = -np.full(sizes[i], mean_rank * (i + 1), np.float32)
        assert np.allclose(out, expect), (rank, i, out[:4], expect[:4])
    comm.barrier()
    if rank == 0:
        print('DIST_OK')


if __name__ == '__main__':
    main()
