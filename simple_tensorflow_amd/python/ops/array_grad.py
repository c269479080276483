"""Gradients for array ops (analog of reference python/ops/array_grad.py)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import RegisterGradient
from simple_tensorflow_amd.python.ops import array_ops, math_ops


@RegisterGradient('Identity')
def _identity_grad(op, grad):
    return grad


@RegisterGradient('Reshape')
def _reshape_grad(op, grad):
    return [array_ops.reshape(grad, array_ops.shape(op.inputs[0])), None]


@RegisterGradient('ExpandDims')
def _expand_dims_grad(op, grad):
    return [array_ops.reshape(grad, array_ops.shape(op.inputs[0])), None]


@RegisterGradient('Squeeze')
def _squeeze_grad(op, grad):
    return array_ops.reshape(grad, array_ops.shape(op.inputs[0]))


@RegisterGradient('Pack')
def _pack_grad(op, grad):
    axis = op.get_attr('axis')
    return array_ops.unstack(grad, num=len(op.inputs), axis=axis)


@RegisterGradient('Unpack')
def _unpack_grad(op, *grads):
    axis = op.get_attr('axis')
    filled = [g if g is not None else array_ops.zeros_like(op.outputs[i])
              for i, g in enumerate(grads)]
    return array_ops.stack(filled, axis=axis)


@RegisterGradient('ConcatV2')
def _concat_v2_grad(op, grad):
    n = len(op.inputs) - 1
    axis_t = op.inputs[-1]
    axis_v = getattr(axis_t, '_const_value', None)
    shapes = [t._shape for t in op.inputs[:-1]]
    if axis_v is None or any(s is None or any(d is None for d in s)
                             for s in shapes):
        raise NotImplementedError('Concat grad needs static shapes (round 1)')
    axis = int(axis_v) % len(shapes[0])
    outs = []
    offset = 0
    for s in shapes:
        begin = [0] * len(s)
        begin[axis] = offset
        size = list(s)
        outs.append(array_ops.slice(grad, begin, size))
        offset += s[axis]
    return outs + [None]


@RegisterGradient('Concat')
def _concat_grad(op, grad):
    raise NotImplementedError('legacy Concat grad')


@RegisterGradient('Split')
def _split_grad(op, *grads):
    filled = [g if g is not None else array_ops.zeros_like(op.outputs[i])
              for i, g in enumerate(grads)]
    axis_v = getattr(op.inputs[0], '_const_value', None)
    axis = int(axis_v) if axis_v is not None else 0
    return [None, array_ops.concat(list(filled), axis)]


@RegisterGradient('Slice')
def _slice_grad(op, grad):
    x = op.inputs[0]
    begin = op.inputs[1]
    bv = getattr(begin, '_const_value', None)
    if x._shape is None or any(d is None for d in x._shape) or bv is None or \
            grad._shape is None or any(d is None for d in grad._shape):
        raise NotImplementedError('Slice grad needs static shapes (round 1)')
    paddings = [[int(b), int(xs) - int(b) - int(gs)]
                for b, xs, gs in zip(bv.reshape(-1), x._shape, grad._shape)]
    return [array_ops.pad(grad, paddings), None, None]


@RegisterGradient('Pad')
def _pad_grad(op, grad):
    pv = getattr(op.inputs[1], '_const_value', None)
    x = op.inputs[0]
    if pv is None or x._shape is None or any(d is None for d in x._shape):
        raise NotImplementedError('Pad grad needs static shapes (round 1)')
    pv = pv.reshape(-1, 2)
    begin = [int(p[0]) for p in pv]
    size = list(x._shape)
    return [array_ops.slice(grad, begin, size), None]


@RegisterGradient('Transpose')
def _transpose_grad(op, grad):
    perm = op.inputs[1]
    pv = getattr(perm, '_const_value', None)
    if pv is None:
        raise NotImplementedError('Transpose grad needs static perm')
    inv = [0] * len(pv.reshape(-1))
    for i, p in enumerate(pv.reshape(-1)):
        inv[int(p)] = i
    return [array_ops.transpose(grad, inv), None]


@RegisterGradient('Tile')
def _tile_grad(op, grad):
    x = op.inputs[0]
    mv = getattr(op.inputs[1], '_const_value', None)
    if mv is None or x._shape is None or any(d is None for d in x._shape):
        raise NotImplementedError('Tile grad needs static shapes')
    # reshape to [m0, s0, m1, s1, ...] and sum over the m axes
    interleaved = []
    sum_axes = []
    for i, (m, s) in enumerate(zip(mv.reshape(-1), x._shape)):
        sum_axes.append(len(interleaved))
        interleaved.append(int(m))
        interleaved.append(int(s))
    g = array_ops.reshape(grad, interleaved)
    out = math_ops.reduce_sum(g, sum_axes)
    out.set_shape(list(x._shape))
    return [out, None]


@RegisterGradient('Gather')
def _gather_grad(op, grad):
    params = op.inputs[0]
    indices = op.inputs[1]
    if params._shape is None or params._shape[0] is None:
        raise NotImplementedError('Gather grad needs static param rows')
    num_rows = params._shape[0]
    flat_idx = array_ops.reshape(indices, [-1])
    gshape = [-1] + list(params._shape[1:])
    flat_grad = array_ops.reshape(grad, gshape)
    out = array_ops.unsorted_segment_sum(flat_grad, flat_idx, num_rows)
    out.set_shape(list(params._shape))
    return [out, None]


@RegisterGradient('Fill')
def _fill_grad(op, grad):
    return [None, math_ops.reduce_sum(grad)]


@RegisterGradient('BiasAdd')
def _bias_add_grad(op, grad):
    from simple_tensorflow_amd.python.framework.ops import apply_op
    dbias = apply_op('BiasAddGrad', grad)
    return grad, dbias


@RegisterGradient('CheckNumerics')
def _check_numerics_grad(op, grad):
    return array_ops.check_numerics(grad, 'grad of CheckNumerics')


@RegisterGradient('Switch')
def _switch_grad(op, *grads):
    # merge the two branch grads back
    from simple_tensorflow_amd.python.ops import control_flow_ops
    g0 = grads[0] if grads[0] is not None else None
    g1 = grads[1] if grads[1] is not None else None
    if g0 is None and g1 is None:
        return None, None
    if g0 is None:
        g0 = array_ops.zeros_like(g1)
    if g1 is None:
        g1 = array_ops.zeros_like(g0)
    m, _ = control_flow_ops.merge([g0, g1])
    return m, None


@RegisterGradient('Merge')
def _merge_grad(op, grad, _index_grad=None):
    # route the grad to the branch that produced the value
    from simple_tensorflow_amd.python.ops import control_flow_ops
    # Approximation (cond-only): send the grad through a Switch on the same
    # predicate. Locate the pred by looking at the Switch feeding input 0.
    src = op.inputs[0].op
    if src.type == 'Switch':
        pred = src.inputs[1]
        gf, gt = control_flow_ops.switch(grad, pred)
        return [gf, gt]
    return [grad] * len(op.inputs)


for _op in ('ZerosLike', 'OnesLike', 'StopGradient', 'Shape', 'ShapeN',
            'UnsortedSegmentSum', 'OneHot', 'InvertPermutation'):
    ops.NoGradient(_op)


@RegisterGradient('Enter')
def _enter_grad(op, grad):
    # constant-capture Enter inside a rematerialized while body: the
    # gradient passes straight through to the captured external tensor.
    return [grad]


@RegisterGradient('Exit')
def _exit_grad(op, grad):
    return [grad]


@RegisterGradient('NextIteration')
def _next_iteration_grad(op, grad):
    return [grad]


# TensorArray reads act as sweep boundaries in the while-loop gradient
# (values are forward-recorded; nothing upstream needs their gradient).
@RegisterGradient('TensorArrayReadV3')
def _ta_read_grad(op, grad):
    return [None, None, None]


@RegisterGradient('TensorArrayWriteV3')
def _ta_write_grad(op, grad):
    return [None, None, None, None]
