"""ConditionalAccumulator + SyncReplicasOptimizer (reference
sync_replicas_optimizer.py:40 semantics in a single process)."""
import numpy as np

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.ops import data_flow_ops, variables


def setup_function(_):
    tf.reset_default_graph()


def test_conditional_accumulator_average():
    acc = data_flow_ops.ConditionalAccumulator(tf.float32, shape=[2])
    a1 = acc.apply_grad(tf.constant([1.0, 2.0]), local_step=0)
    a2 = acc.apply_grad(tf.constant([3.0, 6.0]), local_step=0)
    take = acc.take_grad(2)
    n = acc.num_accumulated()
    with tf.Session() as s:
        s.run([a1])
        assert s.run(n) == 1
        s.run([a2])
        v = s.run(take)
    np.testing.assert_allclose(v, [2.0, 4.0])


def test_stale_gradient_dropped():
    acc = data_flow_ops.ConditionalAccumulator(tf.float32, shape=[1])
    setg = acc.set_global_step(5)
    stale = acc.apply_grad(tf.constant([100.0]), local_step=2)
    fresh1 = acc.apply_grad(tf.constant([2.0]), local_step=7)
    fresh2 = acc.apply_grad(tf.constant([4.0]), local_step=7)
    take = acc.take_grad(2)
    with tf.Session() as s:
        s.run(setg)
        s.run(stale)
        s.run(fresh1)
        s.run(fresh2)
        v = s.run(take)
    np.testing.assert_allclose(v, [3.0])


def test_sync_replicas_applies_averaged_update():
    w = variables.Variable(tf.constant([0.0, 0.0]))
    loss = tf.reduce_sum(w.ref() * tf.constant([1.0, 2.0]))
    gstep = tf.train.get_or_create_global_step()
    base = tf.train.GradientDescentOptimizer(1.0)
    opt = tf.train.SyncReplicasOptimizer(base, replicas_to_aggregate=2,
                                         total_num_replicas=2)
    gv = opt.compute_gradients(loss, var_list=[w])
    update = opt.apply_gradients(gv, global_step=gstep)
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        # one run pushes ONE replica's grads; the executor's accumulators
        # need two applications before the take unblocks — simulate two
        # replicas by a second manual apply into the same accumulators.
        extra = [acc.apply_grad(g, local_step=0)
                 for acc, (g, _) in zip(opt._accumulators, gv)]
        s.run([update] + extra)
        v = s.run(w.ref())
    np.testing.assert_allclose(v, [-1.0, -2.0])
