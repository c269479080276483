"""String ops (csrc/kernels/cpu_strings.cc; reference core/ops/string_ops.cc
analog)."""
import base64

import numpy as np

import simple_tensorflow_amd as tf


def setup_function(_):
    tf.reset_default_graph()


def _run(t):
    with tf.Session() as s:
        return s.run(t)


def test_string_join():
    out = _run(tf.string_join([tf.constant(['a', 'b']),
                               tf.constant(['x', 'y'])], separator='-'))
    assert [v.decode() for v in out] == ['a-x', 'b-y']
    # scalar broadcast
    out = _run(tf.string_join([tf.constant(['a', 'b']), tf.constant('!')]))
    assert [v.decode() for v in out] == ['a!', 'b!']


def test_string_split():
    st = tf.string_split(tf.constant(['hello world', 'a,b', '']), ' ')
    with tf.Session() as s:
        idx, vals, shape = s.run([st.indices, st.values, st.dense_shape])
    assert [v.decode() for v in vals] == ['hello', 'world', 'a,b']
    assert idx.tolist() == [[0, 0], [0, 1], [1, 0]]
    assert shape.tolist() == [3, 2]
    st2 = tf.string_split(tf.constant(['a,b,,c']), ',')
    with tf.Session() as s:
        vals2 = s.run(st2.values)
    assert [v.decode() for v in vals2] == ['a', 'b', 'c']


def test_substr():
    out = _run(tf.substr(tf.constant(['hello', 'world']), 1, 3))
    assert [v.decode() for v in out] == ['ell', 'orl']


def test_hash_bucket():
    out = _run(tf.string_to_hash_bucket_fast(
        tf.constant(['a', 'b', 'a']), 100))
    assert out[0] == out[2]
    assert 0 <= out[0] < 100 and 0 <= out[1] < 100
    strong = _run(tf.string_to_hash_bucket_strong(
        tf.constant(['a', 'b']), 50, key=[1, 2]))
    assert all(0 <= v < 50 for v in strong)


def test_string_to_number():
    out = _run(tf.string_to_number(tf.constant(['1.5', '-2', '3e2'])))
    np.testing.assert_allclose(out, [1.5, -2.0, 300.0])
    ints = _run(tf.string_to_number(tf.constant(['42']), out_type=tf.int32))
    assert ints[0] == 42


def test_reduce_join():
    m = tf.constant([['a', 'b'], ['c', 'd']])
    out = _run(tf.reduce_join(m, axis=1, separator=','))
    assert [v.decode() for v in out] == ['a,b', 'c,d']
    out0 = _run(tf.reduce_join(m, axis=0, separator='-'))
    assert [v.decode() for v in out0] == ['a-c', 'b-d']
    full = _run(tf.reduce_join(m, separator=''))
    assert full.decode() == 'abcd'


def test_base64_roundtrip():
    data = ['hello', '', 'a', 'ab', b'\x00\xff'.decode('latin1')]
    enc = _run(tf.encode_base64(tf.constant(['hello', '', 'a', 'ab'])))
    for raw, e in zip(['hello', '', 'a', 'ab'], enc):
        want = base64.urlsafe_b64encode(raw.encode()).rstrip(b'=')
        assert e == want
    dec = _run(tf.decode_base64(tf.constant([e.decode() for e in enc])))
    assert [v.decode() for v in dec] == ['hello', '', 'a', 'ab']
