"""Round-2 op-breadth wave: numpy-reference tests for Where, Unique, TopK,
Cumsum/Cumprod, segment reductions, ReverseV2, ListDiff, DynamicPartition/
Stitch, GatherNd/ScatterNd, diag family, SpaceToDepth, MirrorPad,
ReverseSequence, Bitcast, scatter variable updates — and their gradients
(the reference's kernel-test strategy)."""
import numpy as np
import pytest

import simple_tensorflow_amd as tf


@pytest.fixture(autouse=True)
def fresh_graph():
    tf.reset_default_graph()
    yield


def _run(t, feed=None):
    with tf.Session() as s:
        return s.run(t, feed_dict=feed or {})


def test_where_coordinates():
    c = np.array([[True, False], [False, True], [True, True]])
    out = _run(tf.where(tf.constant(c)))
    np.testing.assert_array_equal(out, np.argwhere(c))


def test_unique_and_counts():
    x = np.array([4, 2, 4, 7, 2, 2], np.int32)
    y, idx = _run(tf.unique(tf.constant(x)))
    np.testing.assert_array_equal(y, [4, 2, 7])
    np.testing.assert_array_equal(y[idx], x)
    y2, idx2, cnt = _run(tf.unique_with_counts(tf.constant(x)))
    np.testing.assert_array_equal(cnt, [2, 3, 1])


def test_top_k():
    x = np.array([[1.0, 5.0, 3.0, 5.0], [9.0, 2.0, 8.0, 0.0]], np.float32)
    v, i = _run(tf.nn.top_k(tf.constant(x), k=2))
    np.testing.assert_array_equal(v, [[5, 5], [9, 8]])
    np.testing.assert_array_equal(i, [[1, 3], [0, 2]])


def test_cumsum_cumprod():
    x = np.array([[1.0, 2.0, 3.0], [4.0, 5.0, 6.0]], np.float32)
    cs = _run(tf.cumsum(tf.constant(x), axis=1))
    np.testing.assert_allclose(cs, np.cumsum(x, 1))
    cp = _run(tf.cumprod(tf.constant(x), axis=0))
    np.testing.assert_allclose(cp, np.cumprod(x, 0))
    cse = _run(tf.cumsum(tf.constant(x), axis=1, exclusive=True,
                         reverse=True))
    want = np.array([[5.0, 3.0, 0.0], [11.0, 6.0, 0.0]])
    np.testing.assert_allclose(cse, want)


def test_cumsum_cumprod_grads():
    xv = np.array([1.0, 2.0, 3.0, 4.0], np.float32)
    x = tf.constant(xv)
    g1 = tf.gradients(tf.reduce_sum(tf.cumsum(x) *
                                    tf.constant([1.0, 0.0, 0.0, 1.0])),
                      [x])[0]
    np.testing.assert_allclose(_run(g1), [2.0, 1.0, 1.0, 1.0])
    g2 = tf.gradients(tf.reduce_sum(tf.cumprod(x)), [x])[0]
    # d/dx_i sum_j cumprod_j = sum_{j>=i} prod_{k<=j, k!=i} x_k
    want = [1 + 2 + 6 + 24, 1 + 3 + 12, 2 + 8, 6]
    np.testing.assert_allclose(_run(g2), want)


def test_segment_reductions():
    data = np.array([[1.0, 2.0], [3.0, 4.0], [5.0, 6.0], [7.0, 8.0]],
                    np.float32)
    ids = np.array([0, 0, 1, 1], np.int32)
    d, i = tf.constant(data), tf.constant(ids)
    np.testing.assert_allclose(_run(tf.segment_sum(d, i)),
                               [[4, 6], [12, 14]])
    np.testing.assert_allclose(_run(tf.segment_mean(d, i)),
                               [[2, 3], [6, 7]])
    np.testing.assert_allclose(_run(tf.segment_max(d, i)),
                               [[3, 4], [7, 8]])
    np.testing.assert_allclose(_run(tf.segment_min(d, i)),
                               [[1, 2], [5, 6]])
    np.testing.assert_allclose(_run(tf.segment_prod(d, i)),
                               [[3, 8], [35, 48]])


def test_segment_grads():
    data = np.array([[1.0, 5.0], [3.0, 4.0], [5.0, 6.0]], np.float32)
    ids = np.array([0, 0, 1], np.int32)
    d = tf.constant(data)
    i = tf.constant(ids)
    w = tf.constant(np.array([[1.0, 2.0], [3.0, 4.0]], np.float32))
    g = tf.gradients(tf.reduce_sum(tf.segment_sum(d, i) * w), [d])[0]
    np.testing.assert_allclose(_run(g), [[1, 2], [1, 2], [3, 4]])
    gm = tf.gradients(tf.reduce_sum(tf.segment_mean(d, i) * w), [d])[0]
    np.testing.assert_allclose(_run(gm), [[0.5, 1], [0.5, 1], [3, 4]])
    gx = tf.gradients(tf.reduce_sum(tf.segment_max(d, i) * w), [d])[0]
    np.testing.assert_allclose(_run(gx), [[0, 2], [1, 0], [3, 4]])


def test_reverse_v2_and_grad():
    x = np.arange(6, dtype=np.float32).reshape(2, 3)
    t = tf.constant(x)
    y = tf.reverse_v2(t, [1])
    np.testing.assert_allclose(_run(y), x[:, ::-1])
    w = tf.constant(np.arange(6, dtype=np.float32).reshape(2, 3))
    g = tf.gradients(tf.reduce_sum(y * w), [t])[0]
    np.testing.assert_allclose(_run(g), np.arange(6).reshape(2, 3)[:, ::-1])


def test_setdiff1d():
    out, idx = _run(tf.setdiff1d(tf.constant([1, 2, 3, 4, 5]),
                                 tf.constant([2, 4])))
    np.testing.assert_array_equal(out, [1, 3, 5])
    np.testing.assert_array_equal(idx, [0, 2, 4])


def test_dynamic_partition_and_grad():
    data = np.arange(10, dtype=np.float32).reshape(5, 2)
    parts = np.array([0, 1, 0, 1, 1], np.int32)
    d = tf.constant(data)
    outs = tf.dynamic_partition(d, tf.constant(parts), 2)
    r0, r1 = _run(outs)
    np.testing.assert_allclose(r0, data[parts == 0])
    np.testing.assert_allclose(r1, data[parts == 1])
    g = tf.gradients([tf.reduce_sum(outs[0] * 2.0),
                      tf.reduce_sum(outs[1] * 3.0)], [d])[0]
    want = np.broadcast_to(np.where((parts == 0)[:, None], 2.0, 3.0), (5, 2))
    np.testing.assert_allclose(_run(g), want)


def test_gather_nd_scatter_nd_and_grads():
    params = np.arange(12, dtype=np.float32).reshape(3, 4)
    idx = np.array([[0, 1], [2, 3]], np.int32)
    p = tf.constant(params)
    y = tf.gather_nd(p, tf.constant(idx))
    np.testing.assert_allclose(_run(y), [1.0, 11.0])
    g = tf.gradients(tf.reduce_sum(y * tf.constant([2.0, 5.0])), [p])[0]
    want = np.zeros((3, 4), np.float32)
    want[0, 1] = 2.0
    want[2, 3] = 5.0
    np.testing.assert_allclose(_run(g), want)

    sc = tf.scatter_nd(tf.constant(np.array([[1], [0], [1]], np.int32)),
                       tf.constant(np.array([5.0, 2.0, 3.0], np.float32)),
                       [3])
    np.testing.assert_allclose(_run(sc), [2.0, 8.0, 0.0])


def test_diag_family():
    v = np.array([1.0, 2.0, 3.0], np.float32)
    d = _run(tf.diag(tf.constant(v)))
    np.testing.assert_allclose(d, np.diag(v))
    dp = _run(tf.diag_part(tf.constant(np.diag(v))))
    np.testing.assert_allclose(dp, v)
    m = np.arange(24, dtype=np.float32).reshape(2, 3, 4)
    mdp = _run(tf.matrix_diag_part(tf.constant(m)))
    np.testing.assert_allclose(mdp, np.stack([np.diag(m[0][:, :3]),
                                              np.diag(m[1][:, :3])]))
    md = _run(tf.matrix_diag(tf.constant(np.array([[1.0, 2.0]], np.float32))))
    np.testing.assert_allclose(md[0], [[1, 0], [0, 2]])
    sd = _run(tf.matrix_set_diag(
        tf.constant(np.zeros((2, 2), np.float32)),
        tf.constant(np.array([7.0, 8.0], np.float32))))
    np.testing.assert_allclose(sd, [[7, 0], [0, 8]])
    bp = _run(tf.matrix_band_part(tf.constant(np.ones((4, 4), np.float32)),
                                  1, 0))
    np.testing.assert_allclose(bp, np.tril(np.triu(np.ones((4, 4)), -1), 0))


def test_space_depth_roundtrip_and_grad():
    x = np.arange(32, dtype=np.float32).reshape(1, 4, 4, 2)
    t = tf.constant(x)
    y = tf.space_to_depth(t, 2)
    z = tf.depth_to_space(y, 2)
    ry, rz = _run([y, z])
    assert ry.shape == (1, 2, 2, 8)
    np.testing.assert_allclose(rz, x)
    g = tf.gradients(tf.reduce_sum(y * y), [t])[0]
    np.testing.assert_allclose(_run(g), 2 * x)


def test_mirror_pad_modes():
    x = np.array([[1.0, 2.0, 3.0]], np.float32)
    r = _run(tf.mirror_pad(tf.constant(x), [[0, 0], [2, 2]], 'REFLECT'))
    np.testing.assert_allclose(r, [[3, 2, 1, 2, 3, 2, 1]])
    s = _run(tf.mirror_pad(tf.constant(x), [[0, 0], [2, 2]], 'SYMMETRIC'))
    np.testing.assert_allclose(s, [[2, 1, 1, 2, 3, 3, 2]])


def test_reverse_sequence():
    x = np.arange(12, dtype=np.float32).reshape(2, 3, 2)
    lens = np.array([2, 3], np.int64)
    out = _run(tf.reverse_sequence(tf.constant(x), tf.constant(lens),
                                   seq_axis=1, batch_axis=0))
    want = x.copy()
    want[0, :2] = x[0, :2][::-1]
    want[1, :3] = x[1, :3][::-1]
    np.testing.assert_allclose(out, want)


def test_bitcast():
    x = np.array([1.0, 2.0], np.float32)
    out = _run(tf.bitcast(tf.constant(x), tf.int32))
    np.testing.assert_array_equal(out, x.view(np.int32))


def test_scatter_update_mul():
    v = tf.Variable(np.ones((4, 2), np.float32))
    ref = v._as_graph_element()
    from simple_tensorflow_amd.python.framework.ops import apply_op
    upd = apply_op('ScatterUpdate', ref,
                   tf.constant(np.array([1, 3], np.int32)),
                   tf.constant(np.full((2, 2), 7.0, np.float32)))
    mul = apply_op('ScatterMul', ref,
                   tf.constant(np.array([0], np.int32)),
                   tf.constant(np.full((1, 2), 3.0, np.float32)))
    with tf.Session() as s:
        s.run(tf.global_variables_initializer())
        s.run(upd)
        s.run(mul)
        out = s.run(v.value())
    np.testing.assert_allclose(out, [[3, 3], [7, 7], [1, 1], [7, 7]])


def test_strided_slice_full_semantics():
    x = np.arange(120, dtype=np.float32).reshape(2, 3, 4, 5)
    t = tf.constant(x)
    cases = [
        (lambda a: a[1], lambda a: a[1]),
        (lambda a: a[:, 1:3], lambda a: a[:, 1:3]),
        (lambda a: a[:, ::2, 1::2], lambda a: a[:, ::2, 1::2]),
        (lambda a: a[..., 2], lambda a: a[..., 2]),
        (lambda a: a[1, ..., ::-1], lambda a: a[1, ..., ::-1]),
        (lambda a: a[None, 0, -1], lambda a: a[None, 0, -1]),
        (lambda a: a[:, -2:, :, ::-2], lambda a: a[:, -2:, :, ::-2]),
    ]
    for i, (tf_fn, np_fn) in enumerate(cases):
        out = _run(tf_fn(t))
        want = np_fn(x)
        assert out.shape == want.shape, (i, out.shape, want.shape)
        np.testing.assert_allclose(out, want, err_msg=str(i))


def test_strided_slice_grad():
    x = np.arange(24, dtype=np.float32).reshape(4, 6)
    t = tf.constant(x)
    y = t[1:3, ::2]
    w = tf.constant(np.array([[1.0, 2.0, 3.0], [4.0, 5.0, 6.0]], np.float32))
    g = tf.gradients(tf.reduce_sum(y * w), [t])[0]
    out = _run(g)
    want = np.zeros((4, 6), np.float32)
    want[1:3, ::2] = [[1, 2, 3], [4, 5, 6]]
    np.testing.assert_allclose(out, want)


def test_strided_slice_ellipsis_grad():
    x = np.arange(24, dtype=np.float32).reshape(2, 3, 4)
    t = tf.constant(x)
    y = t[..., 1]
    g = tf.gradients(tf.reduce_sum(y), [t])[0]
    out = _run(g)
    want = np.zeros((2, 3, 4), np.float32)
    want[..., 1] = 1
    np.testing.assert_allclose(out, want)


def test_hash_table():
    from simple_tensorflow_amd.python.ops import lookup_ops
    table = lookup_ops.HashTable(
        lookup_ops.KeyValueTensorInitializer(
            tf.constant(np.array([1, 2, 3], np.int64)),
            tf.constant(np.array([10.0, 20.0, 30.0], np.float32))),
        default_value=-1.0)
    out = table.lookup(tf.constant(np.array([2, 5, 1], np.int64)))
    size = table.size()
    with tf.Session() as s:
        s.run(table.init)
        r, n = s.run([out, size])
    np.testing.assert_allclose(r, [20.0, -1.0, 10.0])
    assert n == 3


def test_mutable_hash_table():
    from simple_tensorflow_amd.python.ops import lookup_ops
    table = lookup_ops.MutableHashTable(tf.int64, tf.float32, -1.0)
    ins = table.insert(tf.constant(np.array([7, 8], np.int64)),
                       tf.constant(np.array([70.0, 80.0], np.float32)))
    out = table.lookup(tf.constant(np.array([8, 9], np.int64)))
    with tf.Session() as s:
        s.run(ins)
        r = s.run(out)
        k, v = s.run(table.export())
    np.testing.assert_allclose(r, [80.0, -1.0])
    assert set(k.tolist()) == {7, 8}


def test_stack_ops():
    from simple_tensorflow_amd.python.framework.ops import apply_op
    h = apply_op('Stack', elem_type=tf.float32)
    p1 = apply_op('StackPush', h, tf.constant(1.0))
    p2 = apply_op('StackPush', h, tf.constant(2.0))
    p2.op._add_control_input(p1.op)
    pop = apply_op('StackPop', h, elem_type=tf.float32)
    pop.op._add_control_input(p2.op)
    pop2 = apply_op('StackPop', h, elem_type=tf.float32)
    pop2.op._add_control_input(pop.op)
    with tf.Session() as s:
        a, b = s.run([pop, pop2])
    assert (a, b) == (2.0, 1.0)


def test_barrier_ops():
    from simple_tensorflow_amd.python.framework.ops import apply_op
    h = apply_op('Barrier', component_types=[tf.float32, tf.float32])
    i0 = apply_op('BarrierInsertMany', h,
                  tf.constant([b'k1', b'k2']),
                  tf.constant(np.array([1.0, 2.0], np.float32)),
                  component_index=0)
    i1 = apply_op('BarrierInsertMany', h,
                  tf.constant([b'k1', b'k2']),
                  tf.constant(np.array([10.0, 20.0], np.float32)),
                  component_index=1)
    take = apply_op('BarrierTakeMany', h, tf.constant(2),
                    component_types=[tf.float32, tf.float32])
    take[0].op._add_control_input(i0.op if hasattr(i0, 'op') else i0)
    take[0].op._add_control_input(i1.op if hasattr(i1, 'op') else i1)
    with tf.Session() as s:
        idx, keys, v0, v1 = s.run(list(take))
    assert sorted(list(keys)) == [b'k1', b'k2']
    np.testing.assert_allclose(sorted(v0.tolist() if hasattr(v0, 'tolist') else list(v0)), [1.0, 2.0])
    np.testing.assert_allclose(sorted(v1.tolist() if hasattr(v1, 'tolist') else list(v1)), [10.0, 20.0])


def test_math_breadth_vs_numpy():
    from scipy import special
    x = np.linspace(-0.9, 0.9, 7).astype(np.float32)
    t = tf.constant(x)
    got = _run([tf.tan(t), tf.asin(t), tf.acos(t), tf.atan(t), tf.erf(t),
                tf.erfc(t), tf.expm1(t)])
    for g, w in zip(got, [np.tan(x), np.arcsin(x), np.arccos(x),
                          np.arctan(x), special.erf(x), special.erfc(x),
                          np.expm1(x)]):
        np.testing.assert_allclose(g, w, rtol=1e-5, atol=1e-6)
    xp = np.array([0.5, 1.0, 2.5, 7.0], np.float32)
    got2 = _run([tf.lgamma(tf.constant(xp)), tf.digamma(tf.constant(xp))])
    np.testing.assert_allclose(got2[0], special.gammaln(xp), rtol=1e-5,
                               atol=1e-5)
    np.testing.assert_allclose(got2[1], special.digamma(xp), rtol=1e-4,
                               atol=1e-4)
    m = _run(tf.mod(tf.constant([7.0, -7.0]), tf.constant([3.0, 3.0])))
    np.testing.assert_allclose(m, np.fmod([7.0, -7.0], 3.0))
    ae = _run(tf.approximate_equal(tf.constant([1.0, 1.1]),
                                   tf.constant([1.0000001, 1.0])))
    np.testing.assert_array_equal(ae, [True, False])


def test_math_breadth_grads():
    x = np.array([0.3, -0.2, 0.7], np.float32)
    t = tf.constant(x)
    for fn, dfn in [(tf.tan, lambda v: 1 / np.cos(v) ** 2),
                    (tf.asin, lambda v: 1 / np.sqrt(1 - v * v)),
                    (tf.atan, lambda v: 1 / (1 + v * v)),
                    (tf.erf, lambda v: 2 / np.sqrt(np.pi) * np.exp(-v * v)),
                    (tf.expm1, np.exp)]:
        g = tf.gradients(tf.reduce_sum(fn(t)), [t])[0]
        np.testing.assert_allclose(_run(g), dfn(x), rtol=1e-4, atol=1e-5)
    g = tf.gradients(tf.reduce_sum(tf.nn.softsign(t)), [t])[0]
    np.testing.assert_allclose(_run(g), 1 / (1 + np.abs(x)) ** 2, rtol=1e-5)


def test_as_string_decode_raw():
    s = _run(tf.as_string(tf.constant(np.array([1, 2], np.int32))))
    assert list(s) == [b'1', b'2']
    raw = np.array([1.5, -2.0], np.float32).tobytes()
    out = _run(tf.decode_raw(tf.constant([raw]), tf.float32))
    np.testing.assert_allclose(out, [[1.5, -2.0]])
