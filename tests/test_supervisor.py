"""tf.train.Supervisor lifecycle (reference supervisor.py analog)."""
import os

import numpy as np

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.ops import variables


def test_supervisor_managed_session(tmp_path):
    tf.reset_default_graph()
    gstep = tf.train.create_global_step()
    v = variables.Variable(tf.constant(0.0), name='v')
    inc = tf.group(v.assign_add(1.0),
                   gstep.assign_add(1))
    sv = tf.train.Supervisor(logdir=str(tmp_path), save_model_secs=0)
    with sv.managed_session('') as sess:
        for _ in range(3):
            if sv.should_stop():
                break
            sess.run(inc)
        val = sess.run(v.ref())
    assert val == 3.0
    # checkpoint written on stop
    assert tf.train.latest_checkpoint(str(tmp_path)) is not None

    # restart: restores from checkpoint
    tf.reset_default_graph()
    gstep = tf.train.create_global_step()
    v = variables.Variable(tf.constant(0.0), name='v')
    sv2 = tf.train.Supervisor(logdir=str(tmp_path))
    with sv2.managed_session('') as sess:
        assert sess.run(v.ref()) == 3.0
