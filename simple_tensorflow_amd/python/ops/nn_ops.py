"""NN ops (analog of reference python/ops/nn_ops.py + nn_impl.py)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import apply_op, convert_to_tensor
from simple_tensorflow_amd.python.ops import array_ops, math_ops, random_ops


def _conv_out_dim(in_dim, k, stride, padding):
    if in_dim is None:
        return None
    if padding == 'SAME':
        return (in_dim + stride - 1) // stride
    return (in_dim - k) // stride + 1


def conv2d(input, filter, strides, padding, use_cudnn_on_gpu=None,  # pylint: disable=redefined-builtin
           data_format='NHWC', name=None):
    x = convert_to_tensor(input)
    w = convert_to_tensor(filter)
    t = apply_op('Conv2D', x, w, strides=list(strides), padding=padding,
                 data_format=data_format, name=name)
    if x._shape is not None and w._shape is not None:
        n, h, wd, _ = x._shape
        r, s, _, k = w._shape
        t.set_shape([n, _conv_out_dim(h, r, strides[1], padding),
                     _conv_out_dim(wd, s, strides[2], padding), k])
    return t


def conv2d_backprop_input(input_sizes, filter, out_backprop, strides, padding,
                          data_format='NHWC', name=None):
    return apply_op('Conv2DBackpropInput',
                    convert_to_tensor(input_sizes, dtype=dtypes.int32),
                    filter, out_backprop, strides=list(strides),
                    padding=padding, data_format=data_format, name=name)


def conv2d_backprop_filter(input, filter_sizes, out_backprop, strides, padding,  # pylint: disable=redefined-builtin
                           data_format='NHWC', name=None):
    return apply_op('Conv2DBackpropFilter', input,
                    convert_to_tensor(filter_sizes, dtype=dtypes.int32),
                    out_backprop, strides=list(strides), padding=padding,
                    data_format=data_format, name=name)


def bias_add(value, bias, data_format='NHWC', name=None):
    v = convert_to_tensor(value)
    t = apply_op('BiasAdd', v, convert_to_tensor(bias, dtype=v.dtype),
                 data_format=data_format, name=name)
    t.set_shape(v._shape)
    return t


def relu(x, name=None):
    x = convert_to_tensor(x)
    t = apply_op('Relu', x, name=name)
    t.set_shape(x._shape)
    return t


def relu6(x, name=None):
    return apply_op('Relu6', x, name=name)


def elu(x, name=None):
    return apply_op('Elu', x, name=name)


def softplus(x, name=None):
    return apply_op('Softplus', x, name=name)


def softmax(logits, name=None):
    t = apply_op('Softmax', convert_to_tensor(logits), name=name)
    t.set_shape(t.op.inputs[0]._shape)
    return t


def log_softmax(logits, name=None):
    return apply_op('LogSoftmax', logits, name=name)


def softmax_cross_entropy_with_logits(labels=None, logits=None, name=None,
                                      dim=-1):
    logits = convert_to_tensor(logits)
    labels = convert_to_tensor(labels, dtype=logits.dtype)
    loss, _ = apply_op('SoftmaxCrossEntropyWithLogits', logits, labels,
                       name=name)
    if logits._shape is not None:
        loss.set_shape([logits._shape[0]])
    return loss


def sparse_softmax_cross_entropy_with_logits(labels=None, logits=None,
                                             name=None):
    logits = convert_to_tensor(logits)
    labels = convert_to_tensor(labels, dtype=dtypes.int64) \
        if not isinstance(labels, ops.Tensor) else labels
    loss, _ = apply_op('SparseSoftmaxCrossEntropyWithLogits', logits, labels,
                       name=name)
    if logits._shape is not None:
        loss.set_shape([logits._shape[0]])
    return loss


def max_pool(value, ksize, strides, padding, data_format='NHWC', name=None):
    x = convert_to_tensor(value)
    t = apply_op('MaxPool', x, ksize=list(ksize), strides=list(strides),
                 padding=padding, data_format=data_format, name=name)
    if x._shape is not None:
        n, h, w, c = x._shape
        t.set_shape([n, _conv_out_dim(h, ksize[1], strides[1], padding),
                     _conv_out_dim(w, ksize[2], strides[2], padding), c])
    return t


def avg_pool(value, ksize, strides, padding, data_format='NHWC', name=None):
    x = convert_to_tensor(value)
    t = apply_op('AvgPool', x, ksize=list(ksize), strides=list(strides),
                 padding=padding, data_format=data_format, name=name)
    if x._shape is not None:
        n, h, w, c = x._shape
        t.set_shape([n, _conv_out_dim(h, ksize[1], strides[1], padding),
                     _conv_out_dim(w, ksize[2], strides[2], padding), c])
    return t


def fused_batch_norm(x, scale, offset, mean=None, variance=None,
                     epsilon=0.001, data_format='NHWC', is_training=True,
                     name=None):
    x = convert_to_tensor(x)
    c = x._shape[-1] if x._shape is not None else None
    if mean is None:
        mean = array_ops.zeros([0] if is_training else [c], x.dtype)
    if variance is None:
        variance = array_ops.zeros([0] if is_training else [c], x.dtype)
    y, batch_mean, batch_var, _, _ = apply_op(
        'FusedBatchNorm', x, scale, offset, mean, variance, epsilon=epsilon,
        data_format=data_format, is_training=is_training, name=name)
    y.set_shape(x._shape)
    if c is not None:
        batch_mean.set_shape([c])
        batch_var.set_shape([c])
    return y, batch_mean, batch_var


def batch_normalization(x, mean, variance, offset, scale, variance_epsilon,
                        name=None):
    inv = math_ops.rsqrt(variance + variance_epsilon)
    if scale is not None:
        inv = inv * scale
    out = x * inv + (offset - mean * inv if offset is not None else -mean * inv)
    return out


def moments(x, axes, name=None, keep_dims=False):
    mean = math_ops.reduce_mean(x, axes, keep_dims=True)
    var = math_ops.reduce_mean(math_ops.squared_difference(x, array_ops.stop_gradient(mean)),
                               axes, keep_dims=True)
    if not keep_dims:
        mean = array_ops.squeeze(mean, axes)
        var = array_ops.squeeze(var, axes)
    return mean, var


def l2_loss(t, name=None):
    return math_ops.l2_loss(t, name=name)


def l2_normalize(x, dim, epsilon=1e-12, name=None):
    sq = math_ops.reduce_sum(math_ops.square(x), dim, keep_dims=True)
    return x * math_ops.rsqrt(math_ops.maximum(sq, epsilon))


def dropout(x, keep_prob, noise_shape=None, seed=None, name=None):
    x = convert_to_tensor(x)
    if isinstance(keep_prob, float) and keep_prob == 1.0:
        return x
    shape = noise_shape if noise_shape is not None else array_ops.shape(x)
    rnd = random_ops.random_uniform(shape, seed=seed, dtype=x.dtype
                                    if x.dtype.is_floating else dtypes.float32)
    mask = math_ops.cast(math_ops.less(rnd, keep_prob), x.dtype)
    return math_ops.divide(x, keep_prob) * mask


def xw_plus_b(x, weights, biases, name=None):
    return bias_add(math_ops.matmul(x, weights), biases, name=name)


def embedding_lookup(params, ids, partition_strategy='mod', name=None,
                     validate_indices=True, max_norm=None):
    if isinstance(params, (list, tuple)) and len(params) > 1:
        raise NotImplementedError('sharded embedding_lookup: round 2')
    if isinstance(params, (list, tuple)):
        params = params[0]
    return array_ops.gather(params, ids, name=name)


def in_top_k(predictions, targets, k, name=None):
    return apply_op('InTopK', predictions, targets, k=k, name=name)


def top_k(input, k=1, sorted=True, name=None):  # pylint: disable=redefined-builtin
    from simple_tensorflow_amd.python.framework import dtypes as _dt
    from simple_tensorflow_amd.python.framework.ops import convert_to_tensor
    return apply_op('TopKV2', convert_to_tensor(input),
                    convert_to_tensor(k, dtype=_dt.int32), sorted=sorted,
                    name=name)


def lrn(input, depth_radius=5, bias=1.0, alpha=1.0, beta=0.5, name=None):  # pylint: disable=redefined-builtin
    t = apply_op('LRN', input, depth_radius=depth_radius, bias=bias,
                 alpha=alpha, beta=beta, name=name)
    t.set_shape(t.op.inputs[0]._shape)
    return t


local_response_normalization = lrn


def depthwise_conv2d(input, filter, strides, padding, rate=None, name=None):  # pylint: disable=redefined-builtin
    """Depthwise 2-D convolution (reference nn_impl.depthwise_conv2d;
    kernels in csrc/kernels/hip/depthwise.hip)."""
    x = convert_to_tensor(input)
    w = convert_to_tensor(filter)
    out = apply_op('DepthwiseConv2dNative', x, w,
                   strides=[int(s) for s in strides], padding=padding,
                   name=name)
    if x._shape is not None and w._shape is not None:
        p = _conv_out_dim(x._shape[1], w._shape[0], strides[1], padding)
        q = _conv_out_dim(x._shape[2], w._shape[1], strides[2], padding)
        out.set_shape([x._shape[0], p, q, w._shape[2] * w._shape[3]])
    return out


depthwise_conv2d_native = depthwise_conv2d


def softsign(features, name=None):
    from simple_tensorflow_amd.python.framework.ops import convert_to_tensor
    return apply_op('Softsign', convert_to_tensor(features), name=name)


# ---------------------------------------------------------------------------
# CTC (reference python/ops/ctc_ops.py; kernel csrc/kernels/cpu_ctc.cc)
# ---------------------------------------------------------------------------
def ctc_loss(labels, inputs, sequence_length,
             preprocess_collapse_repeated=False, ctc_merge_repeated=True,
             time_major=True):
    """labels: SparseTensor [batch, time] int32; inputs: [T, B, C] logits
    (blank = C-1); returns loss [B]."""
    if not time_major:
        inputs = array_ops.transpose(inputs, [1, 0, 2])
    loss, _ = apply_op(
        'CTCLoss', inputs, labels.indices,
        math_ops.cast(labels.values, dtypes.int32),
        convert_to_tensor(sequence_length, dtype=dtypes.int32),
        preprocess_collapse_repeated=preprocess_collapse_repeated,
        ctc_merge_repeated=ctc_merge_repeated)
    return loss


def ctc_greedy_decoder(inputs, sequence_length, merge_repeated=True):
    from simple_tensorflow_amd.python.ops import sparse_ops
    idx, vals, shape, logp = apply_op(
        'CTCGreedyDecoder', inputs,
        convert_to_tensor(sequence_length, dtype=dtypes.int32),
        merge_repeated=merge_repeated)
    return [sparse_ops.SparseTensor(idx, vals, shape)], logp


@ops.RegisterGradient('CTCLoss')
def _ctc_loss_grad(op, grad_loss, _):
    # outputs[1] is d(loss_b)/d(inputs); scale rows by the incoming per-batch
    # loss gradient.
    g = op.outputs[1]  # [T, B, C]
    scaled = math_ops.multiply(
        g, array_ops.reshape(grad_loss, [1, -1, 1]))
    return [scaled, None, None, None]


ops.NoGradient('CTCGreedyDecoder')


def sigmoid_cross_entropy_with_logits(labels=None, logits=None, name=None):
    """Elementwise sigmoid cross-entropy (reference nn_impl.py:112
    formulation, numerically stable): max(x,0) - x*z + log1p(exp(-|x|))."""
    x = convert_to_tensor(logits)
    z = convert_to_tensor(labels, dtype=x.dtype)
    return math_ops.add(
        math_ops.maximum(x, array_ops.zeros_like(x)) - x * z,
        math_ops.log1p(math_ops.exp(-math_ops.abs(x))), name=name)
