"""freeze_graph / strip_unused / optimize_for_inference (reference
python/tools/*_test.py analogs)."""
import os

import numpy as np
import pytest

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.framework import graph_util, pbreader
from simple_tensorflow_amd.python.tools import (freeze_graph,
                                                optimize_for_inference_lib,
                                                strip_unused_lib)


def setup_function(_):
    tf.reset_default_graph()


def test_convert_variables_to_constants():
    from simple_tensorflow_amd.python.ops import variables
    x = tf.placeholder(tf.float32, [None, 3], name='x')
    w = variables.Variable(tf.constant([[1.], [2.], [3.]]), name='w')
    tf.matmul(x, w.ref(), name='y')
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        frozen = graph_util.convert_variables_to_constants(
            s, tf.get_default_graph().as_graph_def(), ['y'])
    ops = {n['name']: n['op'] for n in pbreader.parse_graph_def(frozen)}
    assert ops['w'] == 'Const'
    assert all(op not in ('VariableV2', 'Assign') for op in ops.values())

    tf.reset_default_graph()
    (y,) = tf.import_graph_def(frozen, return_elements=['y:0'], name='')
    with tf.Session() as s:
        r = s.run(y, {tf.get_default_graph().get_tensor_by_name('x:0'):
                      np.array([[1., 1., 1.]], dtype=np.float32)})
    np.testing.assert_allclose(r, [[6.0]])


def test_freeze_graph_file_round_trip(tmp_path):
    from simple_tensorflow_amd.python.ops import variables
    x = tf.placeholder(tf.float32, [None, 2], name='x')
    w = variables.Variable(tf.constant([[5.0], [7.0]]), name='w')
    tf.matmul(x, w.ref(), name='y')
    saver = tf.train.Saver()
    graph_path = str(tmp_path / 'g.pb')
    ckpt = str(tmp_path / 'model.ckpt')
    out_path = str(tmp_path / 'frozen.pb')
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        saver.save(s, ckpt)
        with open(graph_path, 'wb') as f:
            f.write(tf.get_default_graph().as_graph_def())
    freeze_graph.freeze_graph(graph_path, ckpt, 'y', out_path)
    assert os.path.exists(out_path)

    tf.reset_default_graph()
    with open(out_path, 'rb') as f:
        (y,) = tf.import_graph_def(f.read(), return_elements=['y:0'],
                                   name='')
    with tf.Session() as s:
        r = s.run(y, {tf.get_default_graph().get_tensor_by_name('x:0'):
                      np.array([[1.0, 2.0]], dtype=np.float32)})
    np.testing.assert_allclose(r, [[19.0]])


def test_strip_unused():
    x = tf.placeholder(tf.float32, [2], name='x')
    pre = tf.add(x, tf.constant([1.0, 1.0]), name='pre')
    tf.multiply(pre, tf.constant(2.0), name='out')
    tf.sqrt(pre, name='side')  # must be stripped
    gd = tf.get_default_graph().as_graph_def()
    stripped = strip_unused_lib.strip_unused(gd, ['pre'], ['out'],
                                             int(tf.float32))
    names = {n['name']: n for n in pbreader.parse_graph_def(stripped)}
    assert 'side' not in names and 'x' not in names
    assert names['pre']['op'] == 'Placeholder'

    tf.reset_default_graph()
    (out,) = tf.import_graph_def(stripped, return_elements=['out:0'],
                                 name='')
    with tf.Session() as s:
        r = s.run(out, {tf.get_default_graph().get_tensor_by_name('pre:0'):
                        np.array([3.0, 4.0], dtype=np.float32)})
    np.testing.assert_allclose(r, [6.0, 8.0])


def test_remove_training_nodes():
    x = tf.placeholder(tf.float32, [2], name='x')
    ident = tf.identity(x, name='ident')
    tf.add(ident, ident, name='out')
    gd = tf.get_default_graph().as_graph_def()
    cleaned = graph_util.remove_training_nodes(gd)
    nodes = {n['name']: n for n in pbreader.parse_graph_def(cleaned)}
    assert 'ident' not in nodes
    assert nodes['out']['input'] == ['x', 'x']


def test_fold_batch_norms():
    np.random.seed(0)
    xv = np.random.randn(1, 4, 4, 3).astype(np.float32)
    wv = np.random.randn(1, 1, 3, 8).astype(np.float32)
    gamma = np.random.rand(8).astype(np.float32) + 0.5
    beta = np.random.randn(8).astype(np.float32)
    mean = np.random.randn(8).astype(np.float32)
    var = np.random.rand(8).astype(np.float32) + 0.5

    x = tf.placeholder(tf.float32, [1, 4, 4, 3], name='x')
    conv = tf.nn.conv2d(x, tf.constant(wv), [1, 1, 1, 1], 'SAME',
                        name='conv')
    y, _, _ = tf.nn.fused_batch_norm(
        conv, tf.constant(gamma), tf.constant(beta), mean=tf.constant(mean),
        variance=tf.constant(var), epsilon=1e-3, is_training=False)
    yname = y.op.name
    with tf.Session() as s:
        want = s.run(y, {x: xv})
        gd = tf.get_default_graph().as_graph_def()

    folded = optimize_for_inference_lib.fold_batch_norms(gd)
    ops = [n['op'] for n in pbreader.parse_graph_def(folded)]
    tf.reset_default_graph()
    (y2,) = tf.import_graph_def(folded, return_elements=[yname + ':0'],
                                name='')
    with tf.Session() as s:
        got = s.run(y2, {tf.get_default_graph().get_tensor_by_name('x:0'):
                         xv})
    np.testing.assert_allclose(got, want, rtol=1e-4, atol=1e-4)


def test_transform_graph_pipeline(tmp_path):
    import subprocess, sys
    x = tf.placeholder(tf.float32, [2], name='x')
    ident = tf.identity(x, name='mid')
    tf.multiply(ident, tf.constant(3.0), name='out')
    tf.sqrt(ident, name='dead')
    in_pb = str(tmp_path / 'in.pb')
    out_pb = str(tmp_path / 'out.pb')
    with open(in_pb, 'wb') as f:
        f.write(tf.get_default_graph().as_graph_def())
    r = subprocess.run(
        [sys.executable, '-m',
         'simple_tensorflow_amd.python.tools.transform_graph',
         '--in_graph', in_pb, '--out_graph', out_pb,
         '--inputs', 'x', '--outputs', 'out',
         '--transforms',
         'remove_nodes(op=Identity) strip_unused_nodes'],
        capture_output=True, text=True)
    assert r.returncode == 0, r.stderr
    with open(out_pb, 'rb') as f:
        names = {n['name']: n for n in pbreader.parse_graph_def(f.read())}
    assert 'dead' not in names and 'mid' not in names

    tf.reset_default_graph()
    with open(out_pb, 'rb') as f:
        (out,) = tf.import_graph_def(f.read(), return_elements=['out:0'],
                                     name='')
    with tf.Session() as s:
        v = s.run(out, {tf.get_default_graph().get_tensor_by_name('x:0'):
                        np.array([1.0, 2.0], dtype=np.float32)})
    np.testing.assert_allclose(v, [3.0, 6.0])


def test_benchmark_model(tmp_path, capsys):
    from simple_tensorflow_amd.python.tools import benchmark_model
    x = tf.placeholder(tf.float32, [4, 8], name='x')
    w = tf.constant(np.random.RandomState(0).randn(8, 8).astype(np.float32))
    tf.matmul(x, w, name='y')
    pb = str(tmp_path / 'g.pb')
    with open(pb, 'wb') as f:
        f.write(tf.get_default_graph().as_graph_def())
    dt, rows = benchmark_model.benchmark(
        pb, [('x', (4, 8), np.float32)], ['y'], num_runs=3, warmup=1)
    assert dt > 0
    assert any(r['name'] == 'MatMul' for r in rows)
