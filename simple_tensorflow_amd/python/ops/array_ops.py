"""Array ops (analog of reference python/ops/array_ops.py)."""
import builtins as _bi
import numpy as np

from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import apply_op, convert_to_tensor


def placeholder(dtype, shape=None, name=None):
    dt = dtypes.as_dtype(dtype)
    t = apply_op('Placeholder', dtype=dt,
                 shape=None if shape is None else list(shape), name=name)
    t.set_shape(shape)
    return t


def identity(x, name=None):
    return apply_op('Identity', x, name=name)


def stop_gradient(x, name=None):
    return apply_op('StopGradient', x, name=name)


def shape(x, name=None, out_type=dtypes.int32):
    x = convert_to_tensor(x)
    t = apply_op('Shape', x, out_type=out_type, name=name)
    if x._shape is not None:
        t.set_shape([len(x._shape)])
        if all(d is not None for d in x._shape):
            t._const_value = np.array(x._shape,
                                      dtype=out_type.as_numpy_dtype)
    return t


def shape_n(xs, name=None):
    return [shape(x) for x in xs]


def rank(x, name=None):
    return apply_op('Rank', x, name=name)


def size(x, name=None):
    return apply_op('Size', x, name=name)


def reshape(x, new_shape, name=None):
    x = convert_to_tensor(x)
    t = apply_op('Reshape', x, convert_to_tensor(new_shape, dtype=dtypes.int32),
                 name=name)
    # static shape
    sv = _static_value(t.op.inputs[1])
    if sv is not None:
        dims = [int(d) for d in sv]
        if -1 in dims:
            known = 1
            for d in dims:
                if d != -1:
                    known *= d
            total = _num_elements(x)
            if total is not None:
                dims[dims.index(-1)] = total // known
            else:
                dims[dims.index(-1)] = None
        t.set_shape(dims)
    return t


def _num_elements(x):
    if x._shape is None or any(d is None for d in x._shape):
        return None
    n = 1
    for d in x._shape:
        n *= d
    return n


def _static_value(t):
    v = getattr(t, '_const_value', None)
    return v


def expand_dims(x, axis, name=None):
    x = convert_to_tensor(x)
    t = apply_op('ExpandDims', x, convert_to_tensor(axis, dtype=dtypes.int32),
                 name=name)
    if x._shape is not None:
        s = list(x._shape)
        a = axis if axis >= 0 else axis + len(s) + 1
        s.insert(a, 1)
        t.set_shape(s)
    return t


def squeeze(x, axis=None, name=None, squeeze_dims=None):
    if squeeze_dims is not None:
        axis = squeeze_dims
    x = convert_to_tensor(x)
    t = apply_op('Squeeze', x, squeeze_dims=list(axis) if axis else [],
                 name=name)
    if x._shape is not None:
        dims = []
        nd = len(x._shape)
        ax = [a % nd for a in (axis or [])]
        for i, d in enumerate(x._shape):
            if (not ax and d == 1) or (ax and i in ax):
                continue
            dims.append(d)
        t.set_shape(dims)
    return t


def zeros(shape_arg, dtype=dtypes.float32, name=None):
    return _filled(shape_arg, 0, dtype, name or 'zeros')


def ones(shape_arg, dtype=dtypes.float32, name=None):
    return _filled(shape_arg, 1, dtype, name or 'ones')


def _filled(shape_arg, value, dtype, name):
    dt = dtypes.as_dtype(dtype)
    if isinstance(shape_arg, ops.Tensor):
        v = ops.constant(value, dtype=dt)
        t = apply_op('Fill', shape_arg, v, name=name)
        sv = _static_value(shape_arg)
        if sv is not None:
            t.set_shape([int(d) for d in sv])
        return t
    dims = [int(d) for d in shape_arg]
    n = 1
    for d in dims:
        n *= d
    if n > 256:
        # Fill op instead of a materialized Const proto.
        t = apply_op('Fill', convert_to_tensor(dims, dtype=dtypes.int32),
                     ops.constant(value, dtype=dt), name=name)
        t.set_shape(dims)
        return t
    return ops.constant(value, dtype=dt, shape=dims, name=name)


def fill(dims, value, name=None):
    return apply_op('Fill', convert_to_tensor(dims, dtype=dtypes.int32),
                    convert_to_tensor(value), name=name)


def zeros_like(x, dtype=None, name=None):
    x = convert_to_tensor(x)
    if dtype is not None and dtypes.as_dtype(dtype) != x.dtype:
        from simple_tensorflow_amd.python.ops import math_ops
        return zeros_like(math_ops.cast(x, dtype), name=name)
    t = apply_op('ZerosLike', x, name=name)
    t.set_shape(x._shape)
    return t


def ones_like(x, dtype=None, name=None):
    x = convert_to_tensor(x)
    t = apply_op('OnesLike', x, name=name)
    t.set_shape(x._shape)
    return t


def concat(values, axis, name=None):
    if isinstance(values, ops.Tensor):
        values = [values]
    values = [convert_to_tensor(v) for v in values]
    t = apply_op('ConcatV2', values, convert_to_tensor(axis, dtype=dtypes.int32),
                 name=name)
    shapes = [v._shape for v in values]
    if all(s is not None for s in shapes):
        nd = len(shapes[0])
        a = axis % nd
        dims = list(shapes[0])
        tot = 0
        for s in shapes:
            if s[a] is None:
                tot = None
                break
            tot += s[a]
        dims[a] = tot
        t.set_shape(dims)
    return t


def split(value, num_or_size_splits, axis=0, name=None):
    value = convert_to_tensor(value)
    if isinstance(num_or_size_splits, int):
        res = apply_op('Split', convert_to_tensor(axis, dtype=dtypes.int32),
                       value, num_split=num_or_size_splits, name=name)
        outs = list(res) if isinstance(res, tuple) else [res]
        if value._shape is not None:
            nd = len(value._shape)
            a = axis % nd
            dims = list(value._shape)
            if dims[a] is not None:
                dims[a] //= num_or_size_splits
            for o in outs:
                o.set_shape(dims)
        return outs
    raise NotImplementedError('size_splits list: use multiple slice ops')


def stack(values, axis=0, name=None):
    values = [convert_to_tensor(v) for v in values]
    t = apply_op('Pack', values, axis=axis, name=name)
    if values[0]._shape is not None:
        dims = list(values[0]._shape)
        dims.insert(axis if axis >= 0 else axis + len(dims) + 1, len(values))
        t.set_shape(dims)
    return t


pack = stack


def unstack(value, num=None, axis=0, name=None):
    value = convert_to_tensor(value)
    if num is None:
        if value._shape is None or value._shape[axis] is None:
            raise ValueError('Cannot infer num from shape')
        num = value._shape[axis]
    res = apply_op('Unpack', value, num=num, axis=axis, name=name)
    outs = list(res) if isinstance(res, tuple) else [res]
    if value._shape is not None:
        dims = [d for i, d in enumerate(value._shape) if i != axis % len(value._shape)]
        for o in outs:
            o.set_shape(dims)
    return outs


unpack = unstack


def slice(input_, begin, size, name=None):  # pylint: disable=redefined-builtin
    t = apply_op('Slice', convert_to_tensor(input_),
                 convert_to_tensor(begin, dtype=dtypes.int32),
                 convert_to_tensor(size, dtype=dtypes.int32), name=name)
    if not isinstance(size, ops.Tensor):
        x = convert_to_tensor(input_)
        dims = list(size)
        if x._shape is not None and not isinstance(begin, ops.Tensor):
            for i, d in enumerate(dims):
                if d == -1 and x._shape[i] is not None:
                    dims[i] = x._shape[i] - begin[i]
        t.set_shape([d if d != -1 else None for d in dims])
    return t


def strided_slice(input_, begin, end, strides=None, begin_mask=0,
                  end_mask=0, ellipsis_mask=0, new_axis_mask=0,
                  shrink_axis_mask=0, name=None):
    """Full-semantics strided slice (reference array_ops.strided_slice)."""
    x = convert_to_tensor(input_)
    if strides is None:
        strides = [1] * len(begin)
    t = apply_op('StridedSlice', x,
                 convert_to_tensor(begin, dtype=dtypes.int32),
                 convert_to_tensor(end, dtype=dtypes.int32),
                 convert_to_tensor(strides, dtype=dtypes.int32),
                 begin_mask=begin_mask, end_mask=end_mask,
                 ellipsis_mask=ellipsis_mask, new_axis_mask=new_axis_mask,
                 shrink_axis_mask=shrink_axis_mask, name=name)
    # Static shape via a zero-stride numpy view (no allocation).
    bs = _static_value(t.op.inputs[1])
    es = _static_value(t.op.inputs[2])
    ss = _static_value(t.op.inputs[3])
    if x._shape is not None and all(d is not None for d in x._shape) and \
            bs is not None and es is not None and ss is not None:
        key = []
        for i in range(len(bs)):
            if (new_axis_mask >> i) & 1:
                key.append(None)
            elif (ellipsis_mask >> i) & 1:
                key.append(Ellipsis)
            elif (shrink_axis_mask >> i) & 1:
                key.append(int(bs[i]))
            else:
                b = None if (begin_mask >> i) & 1 else int(bs[i])
                e = None if (end_mask >> i) & 1 else int(es[i])
                key.append(_bi.slice(b, e, int(ss[i])))
        fake = np.lib.stride_tricks.as_strided(
            np.zeros(1, np.int8), tuple(x._shape), (0,) * len(x._shape))
        t.set_shape(list(fake[tuple(key)].shape))
    return t


def _slice_helper(tensor, key):
    """tensor[...] -> StridedSlice with the reference's mask encoding
    (python/ops/array_ops.py _SliceHelper)."""
    if not isinstance(key, tuple):
        key = (key,)
    begin, end, strides = [], [], []
    begin_mask = end_mask = ellipsis_mask = new_axis_mask = 0
    shrink_axis_mask = 0
    for i, k in enumerate(key):
        if isinstance(k, _bi.slice):
            if k.start is None:
                begin.append(0)
                begin_mask |= 1 << i
            else:
                begin.append(k.start)
            if k.stop is None:
                end.append(0)
                end_mask |= 1 << i
            else:
                end.append(k.stop)
            strides.append(1 if k.step is None else k.step)
        elif k is Ellipsis:
            begin.append(0)
            end.append(0)
            strides.append(1)
            ellipsis_mask |= 1 << i
        elif k is None:
            begin.append(0)
            end.append(0)
            strides.append(1)
            new_axis_mask |= 1 << i
        elif isinstance(k, int) or (isinstance(k, np.integer)):
            begin.append(int(k))
            end.append(int(k) + 1)
            strides.append(1)
            shrink_axis_mask |= 1 << i
        elif isinstance(k, ops.Tensor):
            raise NotImplementedError('tensor slice indices')
        else:
            raise TypeError('invalid slice key %r' % (k,))
    return strided_slice(tensor, begin, end, strides,
                         begin_mask=begin_mask, end_mask=end_mask,
                         ellipsis_mask=ellipsis_mask,
                         new_axis_mask=new_axis_mask,
                         shrink_axis_mask=shrink_axis_mask)


def pad(x, paddings, name=None):
    t = apply_op('Pad', convert_to_tensor(x),
                 convert_to_tensor(paddings, dtype=dtypes.int32), name=name)
    x = convert_to_tensor(x)
    if x._shape is not None and not isinstance(paddings, ops.Tensor):
        dims = []
        for d, (lo, hi) in zip(x._shape, paddings):
            dims.append(None if d is None else d + lo + hi)
        t.set_shape(dims)
    return t


def transpose(x, perm=None, name=None):
    x = convert_to_tensor(x)
    if perm is None:
        nd = len(x._shape)
        perm = list(range(nd))[::-1]
    t = apply_op('Transpose', x, convert_to_tensor(perm, dtype=dtypes.int32),
                 name=name)
    if x._shape is not None:
        if not isinstance(perm, ops.Tensor):
            t.set_shape([x._shape[p] for p in perm])
        else:
            t.set_shape([None] * len(x._shape))  # rank is permutation-invariant
    return t


def gather(params, indices, name=None):
    idx = convert_to_tensor(indices) if hasattr(indices, 'dtype') and \
        getattr(indices, 'dtype', None) in (dtypes.int32, dtypes.int64) \
        else convert_to_tensor(indices, dtype=dtypes.int32)
    if idx.dtype == dtypes.int64:
        from simple_tensorflow_amd.python.ops import math_ops as _mo
        idx = _mo.cast(idx, dtypes.int32)
    t = apply_op('Gather', convert_to_tensor(params), idx, name=name)
    p, i = t.op.inputs
    if p._shape is not None and i._shape is not None:
        t.set_shape(list(i._shape) + list(p._shape[1:]))
    return t


def tile(x, multiples, name=None):
    xt = convert_to_tensor(x)
    mt = convert_to_tensor(multiples, dtype=dtypes.int32)
    t = apply_op('Tile', xt, mt, name=name)
    # Shape inference: per-dim product where both factors are known; rank is
    # known whenever the input rank (or a static multiples vector) is.
    mv = getattr(mt, '_const_value', None)
    if xt._shape is not None:
        if mv is not None:
            t.set_shape([None if d is None else int(d) * int(m)
                         for d, m in zip(xt._shape, mv.reshape(-1))])
        else:
            t.set_shape([None] * len(xt._shape))
    return t


def one_hot(indices, depth, on_value=1.0, off_value=0.0, axis=-1,
            dtype=dtypes.float32, name=None):
    dt = dtypes.as_dtype(dtype)
    t = apply_op('OneHot', convert_to_tensor(indices, dtype=dtypes.int64)
                 if not isinstance(indices, ops.Tensor) else indices,
                 convert_to_tensor(depth, dtype=dtypes.int32),
                 ops.constant(on_value, dtype=dt),
                 ops.constant(off_value, dtype=dt), axis=axis, name=name)
    idx = t.op.inputs[0]
    if idx._shape is not None and isinstance(depth, int):
        t.set_shape(list(idx._shape) + [depth])
    return t


def where(condition, x=None, y=None, name=None):
    if x is None and y is None:
        # coordinate form: [n, rank] int64 indices of true elements
        return apply_op('Where', convert_to_tensor(condition), name=name)
    if (x is None) != (y is None):
        raise ValueError('x and y must both be set or both be None')
    return apply_op('Select', convert_to_tensor(condition),
                    convert_to_tensor(x), convert_to_tensor(y), name=name)


def broadcast_gradient_args(s0, s1, name=None):
    return apply_op('BroadcastGradientArgs', s0, s1, name=name)


def check_numerics(x, message, name=None):
    return apply_op('CheckNumerics', x, message=message, name=name)


def unsorted_segment_sum(data, segment_ids, num_segments, name=None):
    return apply_op('UnsortedSegmentSum', convert_to_tensor(data),
                    convert_to_tensor(segment_ids),
                    convert_to_tensor(num_segments, dtype=dtypes.int32),
                    name=name)


def placeholder_with_default(input, shape=None, name=None):  # pylint: disable=redefined-builtin
    t = apply_op('PlaceholderWithDefault', convert_to_tensor(input),
                 shape=list(shape) if shape is not None else [], name=name)
    if shape is not None:
        t.set_shape(list(shape))
    return t


def eye(num_rows, num_columns=None, dtype=dtypes.float32, name=None):
    m = num_columns if num_columns is not None else num_rows
    arr = np.eye(num_rows, m).astype(dtypes.as_dtype(dtype).as_numpy_dtype)
    return ops.constant(arr, name=name or 'eye')


def meshgrid(*args, **kwargs):
    indexing = kwargs.get('indexing', 'xy')
    tensors = [convert_to_tensor(a) for a in args]
    if len(tensors) != 2:
        raise NotImplementedError('meshgrid supports 2 inputs')
    x, y = tensors
    nx, ny = x._shape[0], y._shape[0]
    if indexing == 'xy':
        X = tile(reshape(x, [1, nx]), [ny, 1])
        Y = tile(reshape(y, [ny, 1]), [1, nx])
    else:
        X = tile(reshape(x, [nx, 1]), [1, ny])
        Y = tile(reshape(y, [1, ny]), [nx, 1])
    return X, Y


def reverse(tensor, axis, name=None):
    t = convert_to_tensor(tensor)
    if isinstance(axis, (list, tuple)):
        for a in axis:
            t = reverse(t, a)
        return t
    a = int(axis) % len(t._shape)
    n = t._shape[a]
    idx = ops.constant(list(range(n - 1, -1, -1)), dtype=dtypes.int32)
    if a == 0:
        return gather(t, idx)
    perm = list(range(len(t._shape)))
    perm[0], perm[a] = perm[a], perm[0]
    return transpose(gather(transpose(t, perm), idx), perm)


reverse_v2 = reverse


# ---------------------------------------------------------------------------
# round-2 breadth wave wrappers
# ---------------------------------------------------------------------------
def unique(x, out_idx=dtypes.int32, name=None):
    return apply_op('Unique', convert_to_tensor(x), name=name)


def unique_with_counts(x, out_idx=dtypes.int32, name=None):
    return apply_op('UniqueWithCounts', convert_to_tensor(x), name=name)


def reverse_v2(tensor, axis, name=None):
    return apply_op('ReverseV2', convert_to_tensor(tensor),
                    convert_to_tensor(axis, dtype=dtypes.int32), name=name)


def setdiff1d(x, y, index_dtype=dtypes.int32, name=None):
    return apply_op('ListDiff', convert_to_tensor(x), convert_to_tensor(y),
                    name=name)


def dynamic_partition(data, partitions, num_partitions, name=None):
    out = apply_op('DynamicPartition', convert_to_tensor(data),
                   convert_to_tensor(partitions, dtype=dtypes.int32),
                   num_partitions=num_partitions, name=name)
    return list(out) if isinstance(out, tuple) else [out]


def gather_nd(params, indices, name=None):
    return apply_op('GatherNd', convert_to_tensor(params),
                    convert_to_tensor(indices, dtype=dtypes.int32), name=name)


def scatter_nd(indices, updates, shape, name=None):
    return apply_op('ScatterNd',
                    convert_to_tensor(indices, dtype=dtypes.int32),
                    convert_to_tensor(updates),
                    convert_to_tensor(shape, dtype=dtypes.int32), name=name)


def diag(diagonal, name=None):
    return apply_op('Diag', convert_to_tensor(diagonal), name=name)


def diag_part(input_, name=None):
    return apply_op('DiagPart', convert_to_tensor(input_), name=name)


def matrix_diag(diagonal, name=None):
    return apply_op('MatrixDiag', convert_to_tensor(diagonal), name=name)


def matrix_diag_part(input_, name=None):
    return apply_op('MatrixDiagPart', convert_to_tensor(input_), name=name)


def matrix_set_diag(input_, diagonal, name=None):
    return apply_op('MatrixSetDiag', convert_to_tensor(input_),
                    convert_to_tensor(diagonal), name=name)


def matrix_band_part(input_, num_lower, num_upper, name=None):
    return apply_op('MatrixBandPart', convert_to_tensor(input_),
                    convert_to_tensor(num_lower, dtype=dtypes.int64),
                    convert_to_tensor(num_upper, dtype=dtypes.int64),
                    name=name)


def space_to_depth(input_, block_size, name=None):
    return apply_op('SpaceToDepth', convert_to_tensor(input_),
                    block_size=block_size, name=name)


def depth_to_space(input_, block_size, name=None):
    return apply_op('DepthToSpace', convert_to_tensor(input_),
                    block_size=block_size, name=name)


def mirror_pad(tensor, paddings, mode, name=None):
    return apply_op('MirrorPad', convert_to_tensor(tensor),
                    convert_to_tensor(paddings, dtype=dtypes.int32),
                    mode=mode, name=name)


def reverse_sequence(input_, seq_lengths, seq_axis=None, batch_axis=None,
                     seq_dim=None, batch_dim=None, name=None):
    seq_dim = seq_axis if seq_axis is not None else seq_dim
    batch_dim = batch_axis if batch_axis is not None else (batch_dim or 0)
    lens = seq_lengths if isinstance(seq_lengths, ops.Tensor) else \
        convert_to_tensor(seq_lengths, dtype=dtypes.int64)
    return apply_op('ReverseSequence', convert_to_tensor(input_), lens,
                    seq_dim=seq_dim, batch_dim=batch_dim, name=name)


def bitcast(input_, type, name=None):  # pylint: disable=redefined-builtin
    return apply_op('Bitcast', convert_to_tensor(input_),
                    type=dtypes.as_dtype(type), name=name)
