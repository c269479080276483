"""tf.losses (reference python/ops/losses/losses_impl.py subset)."""
from simple_tensorflow_amd.python.framework import ops
from simple_tensorflow_amd.python.ops import array_ops, math_ops, nn_ops


class Reduction(object):
    NONE = 'none'
    SUM = 'weighted_sum'
    MEAN = 'weighted_mean'
    SUM_BY_NONZERO_WEIGHTS = 'weighted_sum_by_nonzero_weights'


def _reduce(losses, weights, reduction):
    w = ops.convert_to_tensor(weights, dtype=losses.dtype)
    weighted = losses * w
    if reduction == Reduction.NONE:
        return weighted
    total = math_ops.reduce_sum(weighted)
    if reduction == Reduction.SUM:
        loss = total
    else:
        denom = math_ops.reduce_sum(
            array_ops.ones_like(losses) * w)
        loss = total / denom
    ops.get_default_graph().add_to_collection(ops.GraphKeys.LOSSES, loss)
    return loss


def mean_squared_error(labels, predictions, weights=1.0, scope=None,
                       reduction=Reduction.MEAN):
    with ops.get_default_graph().name_scope(scope or 'mean_squared_error'):
        losses = math_ops.squared_difference(
            ops.convert_to_tensor(predictions),
            ops.convert_to_tensor(labels))
        return _reduce(losses, weights, reduction)


def absolute_difference(labels, predictions, weights=1.0, scope=None,
                        reduction=Reduction.MEAN):
    with ops.get_default_graph().name_scope(scope or 'absolute_difference'):
        losses = math_ops.abs(ops.convert_to_tensor(predictions) -
                              ops.convert_to_tensor(labels))
        return _reduce(losses, weights, reduction)


def sigmoid_cross_entropy(multi_class_labels, logits, weights=1.0,
                          scope=None, reduction=Reduction.MEAN):
    with ops.get_default_graph().name_scope(scope or 'sigmoid_cross_entropy'):
        labels = ops.convert_to_tensor(multi_class_labels,
                                       dtype=logits.dtype)
        # max(x,0) - x*z + log(1 + exp(-|x|))
        zeros = array_ops.zeros_like(logits)
        cond = math_ops.greater_equal(logits, zeros)
        relu_logits = math_ops.select(cond, logits, zeros)
        neg_abs = math_ops.select(cond, math_ops.negative(logits), logits)
        losses = relu_logits - logits * labels + \
            math_ops.log1p(math_ops.exp(neg_abs))
        return _reduce(losses, weights, reduction)


def softmax_cross_entropy(onehot_labels, logits, weights=1.0, scope=None,
                          label_smoothing=0, reduction=Reduction.MEAN):
    with ops.get_default_graph().name_scope(scope or 'softmax_cross_entropy'):
        labels = ops.convert_to_tensor(onehot_labels, dtype=logits.dtype)
        if label_smoothing > 0:
            k = labels._shape[-1]
            labels = labels * (1.0 - label_smoothing) + \
                label_smoothing / float(k)
        losses = nn_ops.softmax_cross_entropy_with_logits(labels=labels,
                                                          logits=logits)
        return _reduce(losses, weights, reduction)


def sparse_softmax_cross_entropy(labels, logits, weights=1.0, scope=None,
                                 reduction=Reduction.MEAN):
    with ops.get_default_graph().name_scope(
            scope or 'sparse_softmax_cross_entropy'):
        losses = nn_ops.sparse_softmax_cross_entropy_with_logits(
            labels=labels, logits=logits)
        return _reduce(losses, weights, reduction)


def hinge_loss(labels, logits, weights=1.0, scope=None,
               reduction=Reduction.MEAN):
    with ops.get_default_graph().name_scope(scope or 'hinge_loss'):
        labels = ops.convert_to_tensor(labels, dtype=logits.dtype)
        all_ones = array_ops.ones_like(labels)
        signs = 2.0 * labels - all_ones
        losses = math_ops.maximum(all_ones - signs * logits,
                                  array_ops.zeros_like(logits))
        return _reduce(losses, weights, reduction)


def log_loss(labels, predictions, weights=1.0, epsilon=1e-7, scope=None,
             reduction=Reduction.MEAN):
    with ops.get_default_graph().name_scope(scope or 'log_loss'):
        labels = ops.convert_to_tensor(labels)
        predictions = ops.convert_to_tensor(predictions)
        losses = -labels * math_ops.log(predictions + epsilon) - \
            (1.0 - labels) * math_ops.log(1.0 - predictions + epsilon)
        return _reduce(losses, weights, reduction)


def get_losses(scope=None):
    return ops.get_default_graph().get_collection(ops.GraphKeys.LOSSES,
                                                  scope)


def get_regularization_losses(scope=None):
    return ops.get_default_graph().get_collection(
        ops.GraphKeys.REGULARIZATION_LOSSES, scope)


def get_total_loss(add_regularization_losses=True, name='total_loss'):
    losses = get_losses()
    if add_regularization_losses:
        losses = losses + get_regularization_losses()
    return math_ops.add_n(losses, name=name)
