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


def _concat_grad_impl(op, grad, inputs, axis_t):
    """Static fast path when every shape is known; otherwise the reference's
    runtime-shape form (array_grad.py:34 _ConcatGrad): ShapeN + ConcatOffset
    feeding Slice with runtime begin/size."""
    axis_v = getattr(axis_t, '_const_value', None)
    shapes = [t._shape for t in inputs]
    if axis_v is not None and not any(
            s is None or any(d is None for d in s) for s in shapes):
        axis = int(axis_v) % len(shapes[0])
        outs = []
        offset = 0
        for s in shapes:
            begin = [0] * len(s)
            begin[axis] = offset
            size = list(s)
            outs.append(array_ops.slice(grad, begin, size))
            offset += s[axis]
        return outs
    from simple_tensorflow_amd.python.framework.ops import apply_op
    sizes = array_ops.shape_n(inputs)
    axis_i32 = math_ops.cast(axis_t, dtypes.int32)
    offsets = apply_op('ConcatOffset', axis_i32, sizes)
    if not isinstance(offsets, (list, tuple)):
        offsets = (offsets,)
    return [array_ops.slice(grad, o, sz) for o, sz in zip(offsets, sizes)]


@RegisterGradient('ConcatV2')
def _concat_v2_grad(op, grad):
    if len(op.inputs) == 2:  # one input + axis: pass-through
        return [grad, None]
    return _concat_grad_impl(op, grad, list(op.inputs[:-1]),
                             op.inputs[-1]) + [None]


@RegisterGradient('Concat')
def _concat_grad(op, grad):
    # legacy form: concat_dim is input 0
    if len(op.inputs) == 2:
        return [None, grad]
    return [None] + _concat_grad_impl(op, grad, list(op.inputs[1:]),
                                      op.inputs[0])


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
    if not (x._shape is None or any(d is None for d in x._shape) or
            bv is None or grad._shape is None or
            any(d is None for d in grad._shape)):
        paddings = [[int(b), int(xs) - int(b) - int(gs)]
                    for b, xs, gs in zip(bv.reshape(-1), x._shape,
                                         grad._shape)]
        return [array_ops.pad(grad, paddings), None, None]
    # Runtime-shape form (reference _SliceGrad): paddings built from
    # shape(x), begin and shape(grad) at execution time.
    begin32 = math_ops.cast(begin, dtypes.int32)
    before = array_ops.reshape(begin32, [-1, 1])
    after = array_ops.reshape(
        array_ops.shape(x) - begin32 - array_ops.shape(grad), [-1, 1])
    paddings = array_ops.concat([before, after], 1)
    return [array_ops.pad(grad, paddings), None, None]


@RegisterGradient('Pad')
def _pad_grad(op, grad):
    pv = getattr(op.inputs[1], '_const_value', None)
    x = op.inputs[0]
    if pv is not None and x._shape is not None and \
            not any(d is None for d in x._shape):
        pv = pv.reshape(-1, 2)
        begin = [int(p[0]) for p in pv]
        size = list(x._shape)
        return [array_ops.slice(grad, begin, size), None]
    # Runtime form: begin = paddings[:, 0], size = shape(x).
    pads = math_ops.cast(op.inputs[1], dtypes.int32)
    begin = array_ops.reshape(
        array_ops.slice(pads, [0, 0], [-1, 1]), [-1])
    return [array_ops.slice(grad, begin, array_ops.shape(x)), None]


@RegisterGradient('Transpose')
def _transpose_grad(op, grad):
    perm = op.inputs[1]
    pv = getattr(perm, '_const_value', None)
    if pv is not None:
        inv = [0] * len(pv.reshape(-1))
        for i, p in enumerate(pv.reshape(-1)):
            inv[int(p)] = i
        return [array_ops.transpose(grad, inv), None]
    from simple_tensorflow_amd.python.framework.ops import apply_op
    return [array_ops.transpose(grad, apply_op('InvertPermutation', perm)),
            None]


@RegisterGradient('Tile')
def _tile_grad(op, grad):
    x = op.inputs[0]
    mv = getattr(op.inputs[1], '_const_value', None)
    if mv is not None and x._shape is not None and \
            not any(d is None for d in x._shape):
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
    # Runtime form (reference _TileGrad): interleave multiples with the
    # input shape; the reduction axes 0,2,4,.. only need the static rank.
    rank = len(x._shape) if x._shape is not None else None
    if rank is None:
        raise NotImplementedError('Tile grad needs a static rank')
    in_shape = array_ops.shape(x)
    mult = math_ops.cast(op.inputs[1], dtypes.int32)
    split_shape = array_ops.reshape(
        array_ops.transpose(array_ops.stack([mult, in_shape])), [-1])
    g = array_ops.reshape(grad, split_shape)
    out = math_ops.reduce_sum(g, list(range(0, 2 * rank, 2)))
    out = array_ops.reshape(out, in_shape)
    if x._shape is not None:
        out.set_shape(list(x._shape))
    return [out, None]


@RegisterGradient('Gather')
def _gather_grad(op, grad):
    params = op.inputs[0]
    indices = op.inputs[1]
    flat_idx = array_ops.reshape(indices, [-1])
    if params._shape is not None and params._shape[0] is not None:
        num_rows = params._shape[0]
        gshape = [-1] + list(params._shape[1:]) \
            if not any(d is None for d in params._shape[1:]) else None
        if gshape is not None:
            flat_grad = array_ops.reshape(grad, gshape)
            out = array_ops.unsorted_segment_sum(flat_grad, flat_idx,
                                                 num_rows)
            out.set_shape(list(params._shape))
            return [out, None]
    # Runtime form: rows from shape(params) at execution time.
    pshape = array_ops.shape(params)
    num_rows = array_ops.reshape(array_ops.slice(pshape, [0], [1]), [])
    inner = array_ops.slice(pshape, [1], [-1])
    gshape = array_ops.concat([ops.constant([-1]), inner], 0)
    flat_grad = array_ops.reshape(grad, gshape)
    out = array_ops.unsorted_segment_sum(flat_grad, flat_idx, num_rows)
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


# TensorArray gradients (reference tensor_array_grad.py design): each
# primary array owns a shadow gradient array (TensorArrayGradV3 resource,
# keyed handle@source). Ordering between the ops touching the shadow array
# rides entirely on the FLOW-gradient scalar chain — a write-grad's read of
# the shadow is data-dependent on the gather-grad's scatter that filled it,
# because the flow gradient it receives descends from that scatter's output
# flow. Works through the while-loop rematerializing backward pass:
# TensorArray writes overwrite (idempotent re-execution), reads don't clear.
def _grad_ta(handle, flow, dtype):
    from simple_tensorflow_amd.python.framework.ops import apply_op
    from simple_tensorflow_amd.python.ops import tensor_array_ops
    g_handle, g_flow = apply_op('TensorArrayGradV3', handle, flow,
                                source='gradients')
    ta = tensor_array_ops.TensorArray(dtype, handle=g_handle, flow=g_flow)
    return ta


@RegisterGradient('TensorArrayReadV3')
def _ta_read_grad(op, grad):
    handle, index, flow = op.inputs
    g = _grad_ta(handle, flow, grad.dtype)
    wflow = g.write(index, grad)._flow
    return [None, None, wflow]


@RegisterGradient('TensorArrayWriteV3')
def _ta_write_grad(op, flow_grad):
    handle, index, value, flow = op.inputs
    if flow_grad is None:
        return [None, None, None, None]
    g = _grad_ta(handle, flow_grad, value.dtype)
    value_grad = g.read(index)
    if value._shape is not None:
        value_grad.set_shape(value._shape)
    return [None, None, value_grad, flow_grad]


@RegisterGradient('TensorArrayGatherV3')
def _ta_gather_grad(op, grad):
    handle, indices, flow = op.inputs
    g = _grad_ta(handle, flow, grad.dtype)
    wflow = g.scatter(indices, grad)._flow
    return [None, None, wflow]


@RegisterGradient('TensorArrayScatterV3')
def _ta_scatter_grad(op, flow_grad):
    handle, indices, value, flow = op.inputs
    if flow_grad is None:
        return [None, None, None, None]
    g = _grad_ta(handle, flow_grad, value.dtype)
    value_grad = g.gather(indices)
    if value._shape is not None:
        value_grad.set_shape(value._shape)
    return [None, None, value_grad, flow_grad]


@RegisterGradient('TensorArrayV3')
def _ta_create_grad(op, *grads):
    return [None]  # size input; the flow chain stops at creation


@RegisterGradient('TensorArrayGradV3')
def _ta_grad_grad(op, *grads):
    return [None, None]


# ---------------------------------------------------------------------------
# round-2 breadth-wave gradients
# ---------------------------------------------------------------------------
@RegisterGradient('GatherNd')
def _gather_nd_grad(op, grad):
    from simple_tensorflow_amd.python.framework.ops import apply_op
    pshape = array_ops.shape(op.inputs[0])
    out = apply_op('ScatterNd', op.inputs[1], grad, pshape)
    return [out, None]


@RegisterGradient('ScatterNd')
def _scatter_nd_grad(op, grad):
    return [None, array_ops.gather_nd(grad, op.inputs[0]), None]


@RegisterGradient('ReverseV2')
def _reverse_v2_grad(op, grad):
    return [array_ops.reverse_v2(grad, op.inputs[1]), None]


@RegisterGradient('ReverseSequence')
def _reverse_sequence_grad(op, grad):
    from simple_tensorflow_amd.python.framework.ops import apply_op
    return [apply_op('ReverseSequence', grad, op.inputs[1],
                     seq_dim=op.get_attr('seq_dim'),
                     batch_dim=op.get_attr('batch_dim')), None]


@RegisterGradient('Diag')
def _diag_grad(op, grad):
    return array_ops.diag_part(grad)


@RegisterGradient('DiagPart')
def _diag_part_grad(op, grad):
    return array_ops.diag(grad)


@RegisterGradient('MatrixDiag')
def _matrix_diag_grad(op, grad):
    return array_ops.matrix_diag_part(grad)


@RegisterGradient('MatrixDiagPart')
def _matrix_diag_part_grad(op, grad):
    return array_ops.matrix_diag(grad)


@RegisterGradient('MatrixSetDiag')
def _matrix_set_diag_grad(op, grad):
    diag_grad = array_ops.matrix_diag_part(grad)
    input_grad = array_ops.matrix_set_diag(
        grad, array_ops.zeros_like(diag_grad))
    return [input_grad, diag_grad]


@RegisterGradient('MatrixBandPart')
def _matrix_band_part_grad(op, grad):
    return [array_ops.matrix_band_part(grad, op.inputs[1], op.inputs[2]),
            None, None]


@RegisterGradient('SpaceToDepth')
def _space_to_depth_grad(op, grad):
    return array_ops.depth_to_space(grad, op.get_attr('block_size'))


@RegisterGradient('DepthToSpace')
def _depth_to_space_grad(op, grad):
    return array_ops.space_to_depth(grad, op.get_attr('block_size'))


@RegisterGradient('DynamicPartition')
def _dynamic_partition_grad(op, *grads):
    """Stitch the per-partition grads back (reference
    data_flow_grad.py _DynamicPartitionGrads)."""
    from simple_tensorflow_amd.python.framework.ops import apply_op
    data = op.inputs[0]
    partitions = op.inputs[1]
    num_partitions = op.get_attr('num_partitions')
    prefix_shape = array_ops.shape(partitions)
    original_indices = array_ops.reshape(
        math_ops.range(0, math_ops.reduce_prod(prefix_shape)), prefix_shape)
    partitioned_indices = array_ops.dynamic_partition(
        original_indices, partitions, num_partitions)
    filled = [g if g is not None else array_ops.zeros_like(op.outputs[i])
              for i, g in enumerate(grads)]
    reconstructed = apply_op('DynamicStitch',
                             list(partitioned_indices), list(filled))
    return [array_ops.reshape(reconstructed, array_ops.shape(data)), None]


for _op in ('Where', 'Unique', 'UniqueWithCounts', 'ListDiff', 'Bitcast',
            'ConcatOffset'):
    ops.NoGradient(_op)


@RegisterGradient('StridedSlice')
def _strided_slice_grad(op, grad):
    from simple_tensorflow_amd.python.framework.ops import apply_op
    out = apply_op('StridedSliceGrad', array_ops.shape(op.inputs[0]),
                   op.inputs[1], op.inputs[2], op.inputs[3], grad,
                   begin_mask=op.get_attr('begin_mask'),
                   end_mask=op.get_attr('end_mask'),
                   ellipsis_mask=op.get_attr('ellipsis_mask'),
                   new_axis_mask=op.get_attr('new_axis_mask'),
                   shrink_axis_mask=op.get_attr('shrink_axis_mask'))
    if op.inputs[0]._shape is not None:
        out.set_shape(list(op.inputs[0]._shape))
    return [out, None, None, None]


@RegisterGradient('StridedSliceGrad')
def _strided_slice_grad_grad(op, grad):
    from simple_tensorflow_amd.python.framework.ops import apply_op
    out = apply_op('StridedSlice', grad, op.inputs[1], op.inputs[2],
                   op.inputs[3],
                   begin_mask=op.get_attr('begin_mask'),
                   end_mask=op.get_attr('end_mask'),
                   ellipsis_mask=op.get_attr('ellipsis_mask'),
                   new_axis_mask=op.get_attr('new_axis_mask'),
                   shrink_axis_mask=op.get_attr('shrink_axis_mask'))
    return [None, None, None, None, out]
