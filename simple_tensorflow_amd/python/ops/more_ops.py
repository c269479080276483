"""Breadth composites (reference nn_impl/array_ops/math_ops scattered
helpers): built on existing ops, no new kernels."""
import numpy as np

from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import convert_to_tensor
from simple_tensorflow_amd.python.ops import (array_ops, math_ops, nn_ops,
                                              random_ops)


def accumulate_n(inputs, shape=None, tensor_dtype=None, name=None):
    return math_ops.add_n(inputs, name=name)


def crelu(features, name=None):
    f = convert_to_tensor(features)
    return array_ops.concat([nn_ops.relu(f), nn_ops.relu(-f)], -1)


def zero_fraction(value, name=None):
    v = convert_to_tensor(value)
    zeros = math_ops.cast(math_ops.equal(v, ops.constant(0, v.dtype)),
                          dtypes.float32)
    return math_ops.reduce_mean(zeros, name=name)


def weighted_cross_entropy_with_logits(targets=None, logits=None,
                                       pos_weight=1.0, name=None):
    x = convert_to_tensor(logits)
    z = convert_to_tensor(targets, dtype=x.dtype)
    w = ops.constant(float(pos_weight), x.dtype)
    one = ops.constant(1.0, x.dtype)
    l = one + (w - one) * z
    # (1-z)x + l*log(1+exp(-|x|)) + l*max(-x,0)
    return (one - z) * x + l * (math_ops.log1p(math_ops.exp(
        -math_ops.abs(x))) + nn_ops.relu(-x))


def conv1d(value, filters, stride, padding, name=None, data_format=None):
    v = array_ops.expand_dims(convert_to_tensor(value), 1)  # [b,1,w,c]
    f = array_ops.expand_dims(convert_to_tensor(filters), 0)
    out = nn_ops.conv2d(v, f, [1, 1, stride, 1], padding)
    return array_ops.squeeze(out, [1])


def separable_conv2d(input, depthwise_filter, pointwise_filter, strides,  # noqa: A002
                     padding, name=None):
    dw = nn_ops.depthwise_conv2d(convert_to_tensor(input),
                                 convert_to_tensor(depthwise_filter),
                                 strides, padding)
    return nn_ops.conv2d(dw, convert_to_tensor(pointwise_filter),
                         [1, 1, 1, 1], 'VALID')


def bincount(arr, weights=None, minlength=None, maxlength=None,
             dtype=dtypes.int32):
    a = math_ops.cast(convert_to_tensor(arr), dtypes.int32)
    flat = array_ops.reshape(a, [-1])
    n = math_ops.reduce_max(flat) + 1
    if minlength is not None:
        n = math_ops.maximum(n, ops.constant(int(minlength), dtypes.int32))
    if weights is None:
        data = array_ops.ones_like(flat)
        data = math_ops.cast(data, dtype)
    else:
        data = array_ops.reshape(convert_to_tensor(weights), [-1])
    return array_ops.unsorted_segment_sum(data, flat, n)


def confusion_matrix(labels, predictions, num_classes=None, weights=None,
                     dtype=dtypes.int32, name=None):
    l = math_ops.cast(convert_to_tensor(labels), dtypes.int32)
    p = math_ops.cast(convert_to_tensor(predictions), dtypes.int32)
    if num_classes is None:
        num_classes = math_ops.maximum(math_ops.reduce_max(l),
                                       math_ops.reduce_max(p)) + 1
        n_t = num_classes
    else:
        n_t = ops.constant(int(num_classes), dtypes.int32)
    idx = l * n_t + p
    if weights is None:
        data = math_ops.cast(array_ops.ones_like(idx), dtype)
    else:
        data = convert_to_tensor(weights)
    flat = array_ops.unsorted_segment_sum(data, idx, n_t * n_t)
    return array_ops.reshape(flat, array_ops.stack([n_t, n_t]) if not
                             isinstance(num_classes, int) else
                             [num_classes, num_classes])


def sufficient_statistics(x, axes, shift=None, keep_dims=False, name=None):
    x = convert_to_tensor(x)
    counts = 1
    for a in axes:
        counts *= int(x._shape[a])
    counts = ops.constant(float(counts), dtypes.float32)
    if shift is not None:
        m = x - shift
    else:
        m = x
    mean_ss = math_ops.reduce_sum(m, axes, keep_dims=keep_dims)
    var_ss = math_ops.reduce_sum(m * m, axes, keep_dims=keep_dims)
    return counts, mean_ss, var_ss, shift


def normalize_moments(counts, mean_ss, variance_ss, shift, name=None):
    divisor = math_ops.reciprocal(counts)
    if shift is not None:
        shifted_mean = mean_ss * divisor
        mean = shifted_mean + shift
    else:
        shifted_mean = mean_ss * divisor
        mean = shifted_mean
    variance = variance_ss * divisor - shifted_mean * shifted_mean
    return mean, variance


def boolean_mask(tensor, mask, name=None):
    t = convert_to_tensor(tensor)
    m = convert_to_tensor(mask)
    idx = array_ops.reshape(array_ops.where(m), [-1])
    idx32 = math_ops.cast(idx, dtypes.int32)
    return array_ops.gather(t, idx32)


def sequence_mask(lengths, maxlen=None, dtype=dtypes.bool, name=None):
    l = math_ops.cast(convert_to_tensor(lengths), dtypes.int32)
    if maxlen is None:
        maxlen = math_ops.reduce_max(l)
    r = math_ops.range(ops.constant(0, dtypes.int32),
                       math_ops.cast(maxlen, dtypes.int32))
    mask = math_ops.less(array_ops.expand_dims(r, 0),
                         array_ops.expand_dims(l, -1))
    if dtype != dtypes.bool:
        mask = math_ops.cast(mask, dtype)
    return mask


def multinomial(logits, num_samples, seed=None, name=None):
    """Gumbel-max sampling: argmax(logits + G) per draw (reference
    multinomial_op.cc semantics, composite formulation)."""
    lg = convert_to_tensor(logits)
    b = int(lg._shape[0])
    c = int(lg._shape[1])
    u = random_ops.random_uniform([b, int(num_samples), c], 1e-20, 1.0,
                                  seed=seed)
    g = -math_ops.log(-math_ops.log(u))
    scores = array_ops.expand_dims(lg, 1) + g
    return math_ops.argmax(scores, axis=2)


def sparse_matmul(a, b, transpose_a=False, transpose_b=False,
                  a_is_sparse=False, b_is_sparse=False, name=None):
    # density hints are an optimization in the reference; dense matmul is
    # numerically identical
    return math_ops.matmul(a, b, transpose_a=transpose_a,
                           transpose_b=transpose_b, name=name)


def space_to_batch(input, paddings, block_size, name=None):  # noqa: A002
    x = array_ops.pad(convert_to_tensor(input), [[0, 0]] +
                      [list(p) for p in np.asarray(paddings)] + [[0, 0]])
    n, h, w, c = [int(d) for d in x._shape]
    bs = int(block_size)
    x = array_ops.reshape(x, [n, h // bs, bs, w // bs, bs, c])
    x = array_ops.transpose(x, [2, 4, 0, 1, 3, 5])
    return array_ops.reshape(x, [n * bs * bs, h // bs, w // bs, c])


def batch_to_space(input, crops, block_size, name=None):  # noqa: A002
    x = convert_to_tensor(input)
    nb, h, w, c = [int(d) for d in x._shape]
    bs = int(block_size)
    n = nb // (bs * bs)
    x = array_ops.reshape(x, [bs, bs, n, h, w, c])
    x = array_ops.transpose(x, [2, 3, 0, 4, 1, 5])
    x = array_ops.reshape(x, [n, h * bs, w * bs, c])
    cr = np.asarray(crops)
    return x[:, int(cr[0][0]):h * bs - int(cr[0][1]),
             int(cr[1][0]):w * bs - int(cr[1][1]), :]


def atrous_conv2d(value, filters, rate, padding, name=None):
    """Dilated conv via space_to_batch / batch_to_space (the reference's own
    lowering, nn_ops.py atrous_conv2d)."""
    if rate == 1:
        return nn_ops.conv2d(value, filters, [1, 1, 1, 1], padding)
    v = convert_to_tensor(value)
    n, h, w, c = [int(d) for d in v._shape]
    fh, fw = int(filters._shape[0]), int(filters._shape[1])
    if padding == 'SAME':
        eff_h, eff_w = fh + (fh - 1) * (rate - 1), fw + (fw - 1) * (rate - 1)
        pad_h, pad_w = eff_h - 1, eff_w - 1
        pt, pl = pad_h // 2, pad_w // 2
        pb, pr = pad_h - pt, pad_w - pl
    else:
        pt = pl = pb = pr = 0
    # pad further to a multiple of rate
    ph = h + pt + pb
    pw_ = w + pl + pr
    eh = (-ph) % rate
    ew = (-pw_) % rate
    sb = space_to_batch(v, [[pt, pb + eh], [pl, pr + ew]], rate)
    out = nn_ops.conv2d(sb, filters, [1, 1, 1, 1], 'VALID')
    return batch_to_space(out, [[0, eh // rate if eh else 0],
                                [0, ew // rate if ew else 0]], rate)


def required_space_to_batch_paddings(input_shape, block_shape,
                                     base_paddings=None, name=None):
    ish = np.asarray(input_shape)
    bsh = np.asarray(block_shape)
    base = np.zeros((len(ish), 2), np.int64) if base_paddings is None \
        else np.asarray(base_paddings)
    pad_start = base[:, 0]
    orig = ish + base.sum(1)
    extra = (-orig) % bsh
    paddings = np.stack([pad_start, base[:, 1] + extra], 1)
    crops = np.stack([np.zeros_like(extra), extra], 1)
    return (ops.constant(paddings.astype(np.int32)),
            ops.constant(crops.astype(np.int32)))


def random_shuffle(value, seed=None, name=None):
    """Shuffle along dim 0 via a random-key top_k permutation (composite;
    reference random_shuffle_op.cc semantics)."""
    from simple_tensorflow_amd.python.ops import nn_ops
    v = convert_to_tensor(value)
    n = int(v._shape[0])
    keys = random_ops.random_uniform([n], seed=seed)
    _, perm = nn_ops.top_k(keys, k=n)
    return array_ops.gather(v, perm)


def random_gamma(shape, alpha, beta=None, dtype=dtypes.float32, seed=None,
                 name=None):
    """Gamma sampling through the py_func bridge (numpy generator; the
    reference uses a Marsaglia-Tsang device kernel)."""
    from simple_tensorflow_amd.python.ops import script_ops
    shape = list(np.asarray(shape).reshape(-1))
    a = float(alpha) if np.isscalar(alpha) else alpha
    b = 1.0 if beta is None else float(beta)
    rng = np.random.RandomState(seed)

    def _draw():
        return (rng.gamma(a, 1.0 / b, size=[int(s) for s in shape])
                .astype(np.float32))

    return script_ops.py_func(_draw, [], dtypes.float32, name=name)


def tables_initializer(name='init_all_tables'):
    from simple_tensorflow_amd.python.ops import control_flow_ops
    g = ops.get_default_graph()
    inits = g.get_collection(ops.GraphKeys.TABLE_INITIALIZERS)
    if not inits:
        return control_flow_ops.no_op(name=name)
    return control_flow_ops.group(*inits, name=name)


def sparse_placeholder(dtype, shape=None, name=None):
    from simple_tensorflow_amd.python.ops import sparse_ops
    idx = array_ops.placeholder(dtypes.int64, [None, None],
                                name=(name or 'sparse') + '/indices')
    vals = array_ops.placeholder(dtype, [None],
                                 name=(name or 'sparse') + '/values')
    shp = array_ops.placeholder(dtypes.int64, [None],
                                name=(name or 'sparse') + '/shape')
    return sparse_ops.SparseTensor(idx, vals, shp)
