"""Math ops (analog of reference python/ops/math_ops.py)."""
import builtins as _bi
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import apply_op, convert_to_tensor


def _bcast_shape(a, b):
    sa, sb = a._shape, b._shape
    if sa is None or sb is None:
        return None
    ra, rb = len(sa), len(sb)
    r = max(ra, rb)
    out = []
    for i in _bi.range(r):
        da = sa[i - r + ra] if i - r + ra >= 0 else 1
        db = sb[i - r + rb] if i - r + rb >= 0 else 1
        if da is None or db is None:
            out.append(None)
        else:
            out.append(max(da, db))
    return out


def _binary(op_name, x, y, name=None):
    x = convert_to_tensor(x) if isinstance(x, ops.Tensor) else x
    if isinstance(x, ops.Tensor) and not isinstance(y, ops.Tensor):
        y = convert_to_tensor(y, dtype=x.dtype)
    elif isinstance(y, ops.Tensor) and not isinstance(x, ops.Tensor):
        x = convert_to_tensor(x, dtype=y.dtype)
    else:
        x = convert_to_tensor(x)
        y = convert_to_tensor(y)
    t = apply_op(op_name, x, y, name=name)
    t.set_shape(_bcast_shape(x, y))
    return t


def add(x, y, name=None): return _binary('Add', x, y, name)
def subtract(x, y, name=None): return _binary('Sub', x, y, name)
sub = subtract
def multiply(x, y, name=None): return _binary('Mul', x, y, name)
mul = multiply
def divide(x, y, name=None): return _binary('RealDiv', x, y, name)
def div(x, y, name=None): return _binary('Div', x, y, name)
truediv = divide
def floordiv(x, y, name=None): return _binary('FloorDiv', x, y, name)
def floormod(x, y, name=None): return _binary('FloorMod', x, y, name)
mod = floormod
def pow(x, y, name=None): return _binary('Pow', x, y, name)  # pylint: disable=redefined-builtin
def maximum(x, y, name=None): return _binary('Maximum', x, y, name)
def minimum(x, y, name=None): return _binary('Minimum', x, y, name)
def squared_difference(x, y, name=None): return _binary('SquaredDifference', x, y, name)
def less(x, y, name=None): return _binary('Less', x, y, name)
def less_equal(x, y, name=None): return _binary('LessEqual', x, y, name)
def greater(x, y, name=None): return _binary('Greater', x, y, name)
def greater_equal(x, y, name=None): return _binary('GreaterEqual', x, y, name)
def equal(x, y, name=None): return _binary('Equal', x, y, name)
def not_equal(x, y, name=None): return _binary('NotEqual', x, y, name)
def logical_and(x, y, name=None): return _binary('LogicalAnd', x, y, name)
def logical_or(x, y, name=None): return _binary('LogicalOr', x, y, name)


def _unary(op_name, x, name=None):
    x = convert_to_tensor(x)
    t = apply_op(op_name, x, name=name)
    t.set_shape(x._shape)
    return t


def negative(x, name=None): return _unary('Neg', x, name)
neg = negative
def abs(x, name=None): return _unary('Abs', x, name)  # pylint: disable=redefined-builtin
def sign(x, name=None): return _unary('Sign', x, name)
def square(x, name=None): return _unary('Square', x, name)
def sqrt(x, name=None): return _unary('Sqrt', x, name)
def rsqrt(x, name=None): return _unary('Rsqrt', x, name)
def exp(x, name=None): return _unary('Exp', x, name)
def log(x, name=None): return _unary('Log', x, name)
def log1p(x, name=None): return _unary('Log1p', x, name)
def tanh(x, name=None): return _unary('Tanh', x, name)
def sigmoid(x, name=None): return _unary('Sigmoid', x, name)
def sin(x, name=None): return _unary('Sin', x, name)
def cos(x, name=None): return _unary('Cos', x, name)
def floor(x, name=None): return _unary('Floor', x, name)
def ceil(x, name=None): return _unary('Ceil', x, name)
def round(x, name=None): return _unary('Round', x, name)  # pylint: disable=redefined-builtin
def reciprocal(x, name=None): return _unary('Reciprocal', x, name)
def logical_not(x, name=None): return _unary('LogicalNot', x, name)
def is_nan(x, name=None): return _unary('IsNan', x, name)
def is_inf(x, name=None): return _unary('IsInf', x, name)
def is_finite(x, name=None): return _unary('IsFinite', x, name)


def cast(x, dtype, name=None):
    x = convert_to_tensor(x)
    dt = dtypes.as_dtype(dtype)
    if x.dtype == dt:
        return x
    t = apply_op('Cast', x, DstT=dt, name=name)
    t.set_shape(x._shape)
    return t


def to_float(x, name=None): return cast(x, dtypes.float32, name)
def to_double(x, name=None): return cast(x, dtypes.float64, name)
def to_int32(x, name=None): return cast(x, dtypes.int32, name)
def to_int64(x, name=None): return cast(x, dtypes.int64, name)


def matmul(a, b, transpose_a=False, transpose_b=False, name=None):
    a = convert_to_tensor(a)
    b = convert_to_tensor(b)
    if a._shape is not None and len(a._shape) > 2:
        t = apply_op('BatchMatMul', a, b, adj_x=transpose_a, adj_y=transpose_b,
                     name=name)
        if a._shape is not None and b._shape is not None:
            m = a._shape[-1] if transpose_a else a._shape[-2]
            n = b._shape[-2] if transpose_b else b._shape[-1]
            t.set_shape(list(a._shape[:-2]) + [m, n])
        return t
    t = apply_op('MatMul', a, b, transpose_a=transpose_a,
                 transpose_b=transpose_b, name=name)
    if a._shape is not None and b._shape is not None:
        m = a._shape[1] if transpose_a else a._shape[0]
        n = b._shape[0] if transpose_b else b._shape[1]
        t.set_shape([m, n])
    return t


def batch_matmul(a, b, adj_x=False, adj_y=False, name=None):
    return apply_op('BatchMatMul', convert_to_tensor(a), convert_to_tensor(b),
                    adj_x=adj_x, adj_y=adj_y, name=name)


def add_n(inputs, name=None):
    inputs = [convert_to_tensor(x) for x in inputs]
    if len(inputs) == 1:
        return inputs[0]
    t = apply_op('AddN', inputs, name=name)
    t.set_shape(inputs[0]._shape)
    return t


def _reduce(op_name, x, axis, keep_dims, name):
    x = convert_to_tensor(x)
    if axis is None:
        if x._shape is not None:
            axis = list(_bi.range(len(x._shape)))
        elif not keep_dims:
            # Unknown rank: full reduction == reduce the flattened vector.
            from simple_tensorflow_amd.python.ops import array_ops
            x = array_ops.reshape(x, [-1])
            axis = [0]
        else:
            raise ValueError('axis=None with keep_dims needs known rank')
    if isinstance(axis, int):
        axis = [axis]
    axis_t = axis if isinstance(axis, ops.Tensor) else \
        convert_to_tensor(list(axis), dtype=dtypes.int32)
    t = apply_op(op_name, x, axis_t, keep_dims=keep_dims, name=name)
    if x._shape is not None and not isinstance(axis, ops.Tensor):
        nd = len(x._shape)
        ax = {a % nd for a in axis}
        dims = []
        for i, d in enumerate(x._shape):
            if i in ax:
                if keep_dims:
                    dims.append(1)
            else:
                dims.append(d)
        t.set_shape(dims)
    return t


def reduce_sum(x, axis=None, keep_dims=False, name=None, reduction_indices=None):
    if reduction_indices is not None:
        axis = reduction_indices
    return _reduce('Sum', x, axis, keep_dims, name)


def reduce_mean(x, axis=None, keep_dims=False, name=None, reduction_indices=None):
    if reduction_indices is not None:
        axis = reduction_indices
    return _reduce('Mean', x, axis, keep_dims, name)


def reduce_max(x, axis=None, keep_dims=False, name=None, reduction_indices=None):
    if reduction_indices is not None:
        axis = reduction_indices
    return _reduce('Max', x, axis, keep_dims, name)


def reduce_min(x, axis=None, keep_dims=False, name=None, reduction_indices=None):
    if reduction_indices is not None:
        axis = reduction_indices
    return _reduce('Min', x, axis, keep_dims, name)


def reduce_prod(x, axis=None, keep_dims=False, name=None, reduction_indices=None):
    if reduction_indices is not None:
        axis = reduction_indices
    return _reduce('Prod', x, axis, keep_dims, name)


def reduce_all(x, axis=None, keep_dims=False, name=None):
    return _reduce('All', x, axis, keep_dims, name)


def reduce_any(x, axis=None, keep_dims=False, name=None):
    return _reduce('Any', x, axis, keep_dims, name)


def _argminmax(op_name, x, axis, output_type, name):
    x = convert_to_tensor(x)
    t = apply_op(op_name, x, convert_to_tensor(axis, dtype=dtypes.int32),
                 output_type=output_type, name=name)
    if x._shape is not None and isinstance(axis, int):
        a = axis % len(x._shape)
        t.set_shape([d for i, d in enumerate(x._shape) if i != a])
    return t


def argmax(x, axis=0, name=None, output_type=dtypes.int64, dimension=None):
    if dimension is not None:
        axis = dimension
    return _argminmax('ArgMax', x, axis, output_type, name)


def argmin(x, axis=0, name=None, output_type=dtypes.int64):
    return _argminmax('ArgMin', x, axis, output_type, name)


def select(cond, x, y, name=None):
    x = convert_to_tensor(x)
    out = apply_op('Select', cond, x, y, name=name)
    if out._shape is None and x._shape is not None:
        out.set_shape(x._shape)
    return out


def range(start, limit=None, delta=1, dtype=None, name=None):  # pylint: disable=redefined-builtin
    if limit is None:
        start, limit = 0, start
    dt = dtypes.as_dtype(dtype) if dtype is not None else dtypes.int32
    return apply_op('Range', convert_to_tensor(start, dtype=dt),
                    convert_to_tensor(limit, dtype=dt),
                    convert_to_tensor(delta, dtype=dt), name=name)


def cumsum(x, axis=0, exclusive=False, reverse=False, name=None):
    return apply_op('Cumsum', convert_to_tensor(x),
                    convert_to_tensor(axis, dtype=dtypes.int32),
                    exclusive=exclusive, reverse=reverse, name=name)


def cumprod(x, axis=0, exclusive=False, reverse=False, name=None):
    return apply_op('Cumprod', convert_to_tensor(x),
                    convert_to_tensor(axis, dtype=dtypes.int32),
                    exclusive=exclusive, reverse=reverse, name=name)


def _segment(op_name, data, segment_ids, name=None):
    return apply_op(op_name, convert_to_tensor(data),
                    convert_to_tensor(segment_ids, dtype=dtypes.int32),
                    name=name)


def segment_sum(data, segment_ids, name=None):
    return _segment('SegmentSum', data, segment_ids, name)


def segment_mean(data, segment_ids, name=None):
    return _segment('SegmentMean', data, segment_ids, name)


def segment_max(data, segment_ids, name=None):
    return _segment('SegmentMax', data, segment_ids, name)


def segment_min(data, segment_ids, name=None):
    return _segment('SegmentMin', data, segment_ids, name)


def segment_prod(data, segment_ids, name=None):
    return _segment('SegmentProd', data, segment_ids, name)


def l2_loss(x, name=None):
    return apply_op('L2Loss', convert_to_tensor(x), name=name)


def global_norm(t_list, name=None):
    from simple_tensorflow_amd.python.ops import array_ops
    halves = [l2_loss(t) for t in t_list if t is not None]
    return sqrt(multiply(add_n(halves), 2.0))


def realdiv(x, y, name=None):
    return _binary('RealDiv', x, y, name)


def div_no_nan(x, y, name=None):
    x = convert_to_tensor(x)
    y = convert_to_tensor(y)
    safe = select(equal(y, ops.constant(0, dtype=y.dtype)),
                  __import__('simple_tensorflow_amd.python.ops.array_ops',
                             fromlist=['x']).ones_like(y), y)
    out = select(equal(y, ops.constant(0, dtype=y.dtype)),
                 __import__('simple_tensorflow_amd.python.ops.array_ops',
                            fromlist=['x']).zeros_like(x), divide(x, safe))
    return out


def norm(tensor, ord='euclidean', axis=None, keep_dims=False, name=None):
    t = convert_to_tensor(tensor)
    if ord in ('euclidean', 2):
        return sqrt(reduce_sum(square(t), axis=axis, keep_dims=keep_dims))
    if ord == 1:
        return reduce_sum(abs(t), axis=axis, keep_dims=keep_dims)
    if ord in ('inf', float('inf')):
        return reduce_max(abs(t), axis=axis, keep_dims=keep_dims)
    raise ValueError('unsupported norm ord %r' % (ord,))


def tensordot(a, b, axes, name=None):
    from simple_tensorflow_amd.python.ops import array_ops
    a = convert_to_tensor(a)
    b = convert_to_tensor(b)
    if isinstance(axes, int):
        a_axes = list(_bi.range(len(a._shape) - axes, len(a._shape)))
        b_axes = list(_bi.range(axes))
    else:
        a_axes, b_axes = axes
        if isinstance(a_axes, int):
            a_axes, b_axes = [a_axes], [b_axes]
    a_free = [i for i in _bi.range(len(a._shape)) if i not in a_axes]
    b_free = [i for i in _bi.range(len(b._shape)) if i not in b_axes]
    ta = array_ops.transpose(a, a_free + list(a_axes))
    tb = array_ops.transpose(b, list(b_axes) + b_free)
    m = 1
    for i in a_free:
        m *= a._shape[i]
    k = 1
    for i in a_axes:
        k *= a._shape[i]
    n = 1
    for i in b_free:
        n *= b._shape[i]
    out = matmul(array_ops.reshape(ta, [m, k]),
                 array_ops.reshape(tb, [k, n]))
    return array_ops.reshape(
        out, [a._shape[i] for i in a_free] + [b._shape[i] for i in b_free])


def trace(x, name=None):
    from simple_tensorflow_amd.python.ops import array_ops
    x = convert_to_tensor(x)
    n = min(x._shape[-2], x._shape[-1])
    idx = ops.constant([[i, i] for i in _bi.range(n)], dtype=dtypes.int64)
    # gather diagonal via reshape trick: flat index i*(cols+1)
    cols = x._shape[-1]
    flat = array_ops.reshape(x, [-1])
    diag_idx = ops.constant([i * (cols + 1) for i in _bi.range(n)],
                            dtype=dtypes.int32)
    return reduce_sum(array_ops.gather(flat, diag_idx))


# ---------------------------------------------------------------------------
# round-2 math breadth
# ---------------------------------------------------------------------------
def tan(x, name=None):
    return _unary('Tan', x, name)


def asin(x, name=None):
    return _unary('Asin', x, name)


def acos(x, name=None):
    return _unary('Acos', x, name)


def atan(x, name=None):
    return _unary('Atan', x, name)


def erf(x, name=None):
    return _unary('Erf', x, name)


def erfc(x, name=None):
    return _unary('Erfc', x, name)


def expm1(x, name=None):
    return _unary('Expm1', x, name)


def lgamma(x, name=None):
    return _unary('Lgamma', x, name)


def digamma(x, name=None):
    return _unary('Digamma', x, name)


def rint(x, name=None):
    return _unary('Rint', x, name)


def softsign(x, name=None):
    return _unary('Softsign', x, name)


def mod(x, y, name=None):
    x = convert_to_tensor(x)
    return apply_op('Mod', x, convert_to_tensor(y, dtype=x.dtype), name=name)


def approximate_equal(x, y, tolerance=1e-5, name=None):
    x = convert_to_tensor(x)
    return apply_op('ApproximateEqual', x,
                    convert_to_tensor(y, dtype=x.dtype),
                    tolerance=tolerance, name=name)
