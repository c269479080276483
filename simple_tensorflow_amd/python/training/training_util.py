"""global_step helpers + learning-rate decay + EMA (analogs of reference
python/training/{training_util,learning_rate_decay,moving_averages}.py)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.ops import (array_ops, control_flow_ops,
                                              math_ops, state_ops, variables)


def create_global_step(graph=None):
    g = graph or ops.get_default_graph()
    if g.get_collection(ops.GraphKeys.GLOBAL_STEP):
        raise ValueError('global step already exists')
    v = variables.Variable(0, dtype=dtypes.int64, trainable=False,
                           name='global_step')
    g.add_to_collection(ops.GraphKeys.GLOBAL_STEP, v)
    return v


def get_global_step(graph=None):
    g = graph or ops.get_default_graph()
    col = g.get_collection(ops.GraphKeys.GLOBAL_STEP)
    return col[0] if col else None


def get_or_create_global_step(graph=None):
    return get_global_step(graph) or create_global_step(graph)


def exponential_decay(learning_rate, global_step, decay_steps, decay_rate,
                      staircase=False, name=None):
    gs = math_ops.cast(global_step._as_graph_element()
                       if hasattr(global_step, '_as_graph_element')
                       else global_step, dtypes.float32)
    p = gs / float(decay_steps)
    if staircase:
        p = math_ops.floor(p)
    return math_ops.multiply(learning_rate,
                             math_ops.pow(float(decay_rate), p), name=name)


def polynomial_decay(learning_rate, global_step, decay_steps,
                     end_learning_rate=0.0001, power=1.0, cycle=False,
                     name=None):
    gs = math_ops.cast(global_step._as_graph_element()
                       if hasattr(global_step, '_as_graph_element')
                       else global_step, dtypes.float32)
    gs = math_ops.minimum(gs, float(decay_steps))
    frac = 1.0 - gs / float(decay_steps)
    return math_ops.add(
        math_ops.multiply(float(learning_rate) - float(end_learning_rate),
                          math_ops.pow(frac, power)),
        float(end_learning_rate), name=name)


def natural_exp_decay(learning_rate, global_step, decay_steps, decay_rate,
                      staircase=False, name=None):
    from simple_tensorflow_amd.python.ops import math_ops
    from simple_tensorflow_amd.python.framework import dtypes, ops as fops
    lr = fops.convert_to_tensor(learning_rate, dtype=dtypes.float32)
    step = math_ops.cast(global_step, dtypes.float32)
    p = step / float(decay_steps)
    if staircase:
        p = math_ops.floor(p)
    return lr * math_ops.exp(fops.constant(-float(decay_rate)) * p)


def inverse_time_decay(learning_rate, global_step, decay_steps, decay_rate,
                       staircase=False, name=None):
    from simple_tensorflow_amd.python.ops import math_ops
    from simple_tensorflow_amd.python.framework import dtypes, ops as fops
    lr = fops.convert_to_tensor(learning_rate, dtype=dtypes.float32)
    step = math_ops.cast(global_step, dtypes.float32)
    p = step / float(decay_steps)
    if staircase:
        p = math_ops.floor(p)
    denom = fops.constant(1.0) + fops.constant(float(decay_rate)) * p
    return lr * math_ops.reciprocal(denom)


def piecewise_constant(x, boundaries, values, name=None):
    xf = math_ops.cast(x._as_graph_element()
                       if hasattr(x, '_as_graph_element') else x,
                       dtypes.float32)
    out = ops.constant(values[-1])
    # build from the right: select first interval that matches
    for b, v in zip(reversed(boundaries), reversed(values[:-1])):
        out = math_ops.select(math_ops.less_equal(xf, float(b)),
                              ops.constant(v), out)
    return out


class ExponentialMovingAverage(object):
    """EMA of variables (analog of python/training/moving_averages.py)."""

    def __init__(self, decay, num_updates=None, name='ExponentialMovingAverage'):
        self._decay = decay
        self._name = name
        self._averages = {}

    def apply(self, var_list):
        g = ops.get_default_graph()
        update_ops = []
        for v in var_list:
            key = v.name
            if key not in self._averages:
                self._averages[key] = variables.Variable(
                    v.value() if hasattr(v, 'value') else v, trainable=False,
                    name=v.name.split(':')[0] + '/' + self._name)
                g.add_to_collection(ops.GraphKeys.MOVING_AVERAGE_VARIABLES,
                                    self._averages[key])
            avg = self._averages[key]
            val = v.value() if hasattr(v, 'value') else v
            update_ops.append(
                state_ops.assign_sub(
                    avg._as_graph_element(),
                    (avg.value() - val) * (1.0 - self._decay)).op)
        return control_flow_ops.group(*update_ops)

    def average(self, var):
        return self._averages.get(var.name)

    def average_name(self, var):
        return var.name.split(':')[0] + '/' + self._name
