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
    comm = dist.init(world, rank)
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
    # average grad over ranks 1..world: mean(rank+1) = 1.5 for world=2
    expect = -np.array([1.5, 3.0])
    assert np.allclose(v, expect), (rank, v)
    comm.barrier()
    if rank == 0:
        print('DIST_OK')


if __name__ == '__main__':
    main()
