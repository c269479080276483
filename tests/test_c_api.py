"""C API (libstf_c.so): compile tests/c/c_api_smoke.c with gcc, run it
against a GraphDef + SavedModel produced here (reference c/c_api_test.cc
analog)."""
import os
import subprocess

import numpy as np
import pytest

import simple_tensorflow_amd as tf

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CLIB = os.path.join(REPO, 'simple_tensorflow_amd', 'libstf_c.so')


@pytest.mark.skipif(not os.path.exists(CLIB), reason='libstf_c.so not built')
def test_c_api_smoke(tmp_path):
    from simple_tensorflow_amd.python.ops import variables
    from simple_tensorflow_amd.python import saved_model as sm

    tf.reset_default_graph()
    x = tf.placeholder(tf.float32, [None, 2], name='x')
    w = variables.Variable(tf.constant([[3.0], [4.0]]), name='w')
    y = tf.matmul(x, w.ref(), name='y')

    graph_pb = str(tmp_path / 'graph.pb')
    export_dir = str(tmp_path / 'sm')
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        frozen = __import__(
            'simple_tensorflow_amd.python.framework.graph_util',
            fromlist=['x']).convert_variables_to_constants(
                s, tf.get_default_graph().as_graph_def(), ['y'])
        with open(graph_pb, 'wb') as f:
            f.write(frozen)
        b = sm.SavedModelBuilder(export_dir)
        b.add_meta_graph_and_variables(s, [sm.tag_constants.SERVING])
        b.save()

    exe = str(tmp_path / 'c_api_smoke')
    csrc_inc = os.path.join(REPO, 'simple_tensorflow_amd', 'csrc')
    compile_cmd = [
        'gcc', '-o', exe, os.path.join(REPO, 'tests', 'c', 'c_api_smoke.c'),
        '-I', csrc_inc, '-L', os.path.dirname(CLIB), '-lstf_c',
        '-Wl,-rpath,' + os.path.dirname(CLIB),
        '-Wl,-rpath,/opt/rocm/lib']
    subprocess.run(compile_cmd, check=True, capture_output=True)
    r = subprocess.run([exe, graph_pb, export_dir], capture_output=True,
                       text=True, timeout=120)
    assert r.returncode == 0, r.stderr + r.stdout
    assert 'C_API_OK' in r.stdout
