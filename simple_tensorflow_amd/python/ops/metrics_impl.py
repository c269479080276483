"""tf.metrics (reference python/ops/metrics_impl.py subset): streaming
metrics as (value, update_op) pairs over local variables."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.ops import (array_ops, math_ops, state_ops,
                                              variables)


def _metric_var(name, shape=()):
    from simple_tensorflow_amd.python.ops import array_ops as ao
    return variables.Variable(
        ao.zeros(list(shape), dtypes.float32), trainable=False,
        collections=[ops.GraphKeys.LOCAL_VARIABLES], name=name)


def mean(values, weights=None, name=None):
    g = ops.get_default_graph()
    with g.name_scope(name or 'mean'):
        values = ops.convert_to_tensor(values)
        total = _metric_var('total')
        count = _metric_var('count')
        num = math_ops.cast(array_ops.size(values), dtypes.float32)
        if weights is not None:
            w = math_ops.cast(ops.convert_to_tensor(weights), values.dtype)
            values = values * w
            num = math_ops.reduce_sum(
                math_ops.cast(w, dtypes.float32) *
                math_ops.cast(array_ops.ones_like(values), dtypes.float32))
        upd_total = state_ops.assign_add(
            total.ref(), math_ops.cast(math_ops.reduce_sum(values),
                                       dtypes.float32))
        upd_count = state_ops.assign_add(count.ref(), num)
        value = total.ref() / math_ops.maximum(
            count.ref(), ops.constant(1e-12))
        update_op = upd_total / math_ops.maximum(upd_count,
                                                 ops.constant(1e-12))
        return value, update_op


def accuracy(labels, predictions, weights=None, name=None):
    g = ops.get_default_graph()
    with g.name_scope(name or 'accuracy'):
        labels = ops.convert_to_tensor(labels)
        predictions = ops.convert_to_tensor(predictions)
        if predictions.dtype != labels.dtype:
            predictions = math_ops.cast(predictions, labels.dtype)
        correct = math_ops.cast(math_ops.equal(labels, predictions),
                                dtypes.float32)
        return mean(correct, weights=weights, name='acc_mean')


def mean_squared_error(labels, predictions, weights=None, name=None):
    g = ops.get_default_graph()
    with g.name_scope(name or 'mse'):
        se = math_ops.squared_difference(
            ops.convert_to_tensor(predictions),
            ops.convert_to_tensor(labels))
        return mean(se, weights=weights, name='mse_mean')
