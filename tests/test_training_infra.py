"""Queues, Coordinator/QueueRunner, input pipeline, MonitoredSession, hooks,
summaries/TFEvents (SURVEY.md §2.4/§5 capability tests)."""
import glob
import os
import struct

import numpy as np
import pytest

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.ops import data_flow_ops


@pytest.fixture(autouse=True)
def fresh_graph():
    tf.reset_default_graph()
    yield


def test_fifo_queue_basic():
    q = data_flow_ops.FIFOQueue(10, [tf.float32], shapes=[[2]])
    enq = q.enqueue([tf.constant([1.0, 2.0])])
    deq = q.dequeue()
    with tf.Session() as s:
        s.run(enq)
        s.run(enq)
        assert s.run(q.size()) == 2
        np.testing.assert_allclose(s.run(deq), [1.0, 2.0])


def test_queue_dequeue_many_and_close():
    q = data_flow_ops.FIFOQueue(10, [tf.int32], shapes=[[]])
    enq = q.enqueue_many([tf.constant([1, 2, 3, 4, 5])])
    deq = q.dequeue_many(3)
    with tf.Session() as s:
        s.run(enq)
        np.testing.assert_array_equal(s.run(deq), [1, 2, 3])
        s.run(q.close())
        with pytest.raises(tf.errors.OutOfRangeError):
            s.run(q.dequeue_many(5))


def test_random_shuffle_queue():
    q = data_flow_ops.RandomShuffleQueue(20, 0, [tf.int32], shapes=[[]],
                                         seed=7)
    enq = q.enqueue_many([tf.constant(list(range(10)))])
    with tf.Session() as s:
        s.run(enq)
        s.run(q.close())
        got = [int(s.run(q.dequeue())) for _ in range(10)]
    assert sorted(got) == list(range(10))


def test_queue_runner_batch_pipeline():
    data = tf.constant(np.arange(20, dtype=np.float32).reshape(10, 2))
    q = data_flow_ops.FIFOQueue(30, [tf.float32], shapes=[[2]])
    enq = q.enqueue_many([data])
    tf.train.add_queue_runner(tf.train.QueueRunner(q, [enq],
                                                   close_op=q.close()))
    batch = q.dequeue_many(4)
    with tf.Session() as s:
        coord = tf.train.Coordinator()
        threads = tf.train.start_queue_runners(s, coord)
        out = s.run(batch)
        assert out.shape == (4, 2)
        coord.request_stop()
        coord.join(threads, stop_grace_period_secs=5)


def test_monitored_session_with_hooks(tmp_path):
    v = tf.Variable(0.0)
    inc = tf.assign_add(v._as_graph_element(), 1.0)
    hook = tf.train.StopAtStepHook(num_steps=5)
    with tf.train.MonitoredTrainingSession(
            checkpoint_dir=str(tmp_path), hooks=[hook]) as sess:
        n = 0
        while not sess.should_stop():
            sess.run(inc)
            n += 1
            assert n < 50
    assert n == 5
    # checkpoint written at end
    assert tf.train.latest_checkpoint(str(tmp_path)) is not None


def test_monitored_session_restores(tmp_path):
    v = tf.Variable(0.0, name='ctr')
    inc = tf.assign_add(v._as_graph_element(), 1.0)
    with tf.train.MonitoredTrainingSession(checkpoint_dir=str(tmp_path)) as s:
        for _ in range(3):
            s.run(inc)
    tf.reset_default_graph()
    v2 = tf.Variable(0.0, name='ctr')
    with tf.train.MonitoredTrainingSession(checkpoint_dir=str(tmp_path)) as s:
        assert s.run(v2.value()) == 3.0


def test_summaries_and_events_file(tmp_path):
    v = tf.Variable(2.5)
    s_op = tf.summary.scalar('my_metric', v.value())
    h_op = tf.summary.histogram('weights', tf.constant([1.0, 2.0, 2.0, 3.0]))
    merged = tf.summary.merge_all()
    with tf.Session() as sess:
        sess.run(tf.global_variables_initializer())
        data = sess.run(merged)
    w = tf.summary.FileWriter(str(tmp_path))
    w.add_summary(data, global_step=7)
    w.close()
    files = glob.glob(str(tmp_path / 'events.out.tfevents.*'))
    assert len(files) == 1
    from simple_tensorflow_amd.python.lib.io import tf_record
    records = list(tf_record.tf_record_iterator(files[0]))
    assert len(records) == 2  # file_version event + our summary event
    assert b'my_metric' in records[1]
    assert b'weights' in records[1]


def test_tf_record_roundtrip(tmp_path):
    path = str(tmp_path / 'data.tfrecord')
    msgs = [b'hello', b'world', b'x' * 1000]
    with tf.python_io.TFRecordWriter(path) as w:
        for m in msgs:
            w.write(m)
    from simple_tensorflow_amd.python.lib.io import tf_record
    assert list(tf_record.tf_record_iterator(path)) == msgs
