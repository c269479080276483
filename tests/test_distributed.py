"""In-process grpc cluster tests (the reference strategy: real gRPC servers on
localhost — SURVEY.md §4; no fake transports)."""
import numpy as np
import pytest

import simple_tensorflow_amd as tf


@pytest.fixture(autouse=True)
def fresh_graph():
    tf.reset_default_graph()
    yield


def test_local_server_session_roundtrip():
    server = tf.train.Server.create_local_server()
    try:
        a = tf.constant([[1.0, 2.0]])
        b = tf.constant([[3.0], [4.0]])
        c = tf.matmul(a, b)
        with tf.Session(server.target) as s:
            out = s.run(c)
        np.testing.assert_allclose(out, [[11.0]])
    finally:
        server.stop()


def test_remote_variables_persist():
    server = tf.train.Server.create_local_server()
    try:
        v = tf.Variable(1.0)
        inc = tf.assign_add(v._as_graph_element(), 1.0)
        with tf.Session(server.target) as s:
            s.run(tf.global_variables_initializer())
            for _ in range(3):
                s.run(inc)
            assert s.run(v.value()) == 4.0
    finally:
        server.stop()


def test_remote_feed_fetch():
    server = tf.train.Server.create_local_server()
    try:
        x = tf.placeholder(tf.float32, [2])
        y = x * 3.0
        with tf.Session(server.target) as s:
            out = s.run(y, feed_dict={x: np.array([1.0, 2.0], np.float32)})
        np.testing.assert_allclose(out, [3.0, 6.0])
    finally:
        server.stop()


def test_cluster_spec():
    cs = tf.train.ClusterSpec({'ps': ['h1:2222'], 'worker': ['h2:2222',
                                                             'h3:2222']})
    assert cs.num_tasks('worker') == 2
    assert cs.task_address('ps', 0) == 'h1:2222'
    setter = tf.train.replica_device_setter(cluster=cs)
    class FakeOp:
        type = 'VariableV2'
    assert setter(FakeOp()).startswith('/job:ps/task:')


def test_data_parallel_two_process_gloo():
    """The RCCL data-parallel graph on 2 CPU processes over the gloo-backed
    collective fallback (the same graph the 8-GPU bench runs)."""
    import subprocess, sys, os
    env = dict(os.environ)
    env['MASTER_ADDR'] = '127.0.0.1'
    env['MASTER_PORT'] = '29713'
    r = subprocess.run(
        [sys.executable, '-m', 'torch.distributed.run', '--nnodes=1',
         '--nproc-per-node', '2', '--master-addr', '127.0.0.1',
         '--master-port', '29713',
         os.path.join(os.path.dirname(__file__), 'helpers',
                      'dist_worker.py')],
        capture_output=True, text=True, timeout=240, env=env)
    assert r.returncode == 0, r.stdout[-2000:] + r.stderr[-2000:]
    assert 'DIST_OK' in r.stdout
