"""Gradient clipping (analog of reference python/ops/clip_ops.py)."""
from simple_tensorflow_amd.python.framework.ops import convert_to_tensor
from simple_tensorflow_amd.python.ops import math_ops


def clip_by_value(t, clip_value_min, clip_value_max, name=None):
    return math_ops.minimum(math_ops.maximum(t, clip_value_min),
                            clip_value_max, name=name)


def clip_by_norm(t, clip_norm, name=None):
    l2 = math_ops.sqrt(math_ops.reduce_sum(math_ops.square(t)))
    factor = math_ops.minimum(1.0, clip_norm / math_ops.maximum(l2, 1e-12))
    return math_ops.multiply(t, factor, name=name)


def global_norm(t_list, name=None):
    halves = [math_ops.l2_loss(t) for t in t_list if t is not None]
    return math_ops.sqrt(math_ops.multiply(math_ops.add_n(halves), 2.0),
                         name=name)


def clip_by_global_norm(t_list, clip_norm, use_norm=None, name=None):
    norm = use_norm if use_norm is not None else global_norm(t_list)
    scale = clip_norm / math_ops.maximum(norm, clip_norm)
    clipped = [None if t is None else math_ops.multiply(t, scale)
               for t in t_list]
    return clipped, norm
