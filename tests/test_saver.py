"""Checkpoint save/restore through SaveV2/RestoreV2 + tensor bundle
(reference format: .index LevelDB table + .data-00000-of-00001 shard)."""
import os

import numpy as np
import pytest

import simple_tensorflow_amd as tf


@pytest.fixture(autouse=True)
def fresh_graph():
    tf.reset_default_graph()
    yield


def test_save_restore_roundtrip(tmp_path):
    v1 = tf.Variable(np.arange(12, dtype=np.float32).reshape(3, 4), name='v1')
    v2 = tf.Variable(np.array([7], dtype=np.int64), name='v2')
    saver = tf.train.Saver()
    prefix = str(tmp_path / 'ckpt')
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        path = saver.save(s, prefix, global_step=3)
        assert path == prefix + '-3'
        assert os.path.exists(path + '.index')
        assert os.path.exists(path + '.data-00000-of-00001')
        # clobber and restore
        s.run(v1.assign(np.zeros((3, 4), np.float32)))
        s.run(v2.assign(np.array([0], np.int64)))
        saver.restore(s, path)
        np.testing.assert_allclose(s.run(v1.value()),
                                   np.arange(12).reshape(3, 4))
        assert s.run(v2.value())[0] == 7
    assert tf.train.latest_checkpoint(str(tmp_path)) == path


def test_restore_into_fresh_session(tmp_path):
    prefix = str(tmp_path / 'm')
    v = tf.Variable(np.array([1.5, -2.5], np.float32), name='w')
    saver = tf.train.Saver()
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        saver.save(s, prefix)
    # fresh graph + session, restore without running the initializer
    tf.reset_default_graph()
    v2 = tf.Variable(np.zeros(2, np.float32), name='w')
    saver2 = tf.train.Saver()
    with tf.Session() as s2:
        saver2.restore(s2, prefix)
        np.testing.assert_allclose(s2.run(v2.value()), [1.5, -2.5])


def test_max_to_keep(tmp_path):
    v = tf.Variable(1.0)
    saver = tf.train.Saver(max_to_keep=2)
    prefix = str(tmp_path / 'k')
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        for i in range(4):
            saver.save(s, prefix, global_step=i)
    assert not os.path.exists(prefix + '-0.index')
    assert not os.path.exists(prefix + '-1.index')
    assert os.path.exists(prefix + '-2.index')
    assert os.path.exists(prefix + '-3.index')
