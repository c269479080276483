"""Initializers (analog of reference python/ops/init_ops.py)."""
import math

from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.ops import array_ops, random_ops


def zeros_initializer(dtype=dtypes.float32):
    def init(shape, dtype_arg=None, partition_info=None):
        return array_ops.zeros(shape, dtype_arg or dtype)
    return init


def ones_initializer(dtype=dtypes.float32):
    def init(shape, dtype_arg=None, partition_info=None):
        return array_ops.ones(shape, dtype_arg or dtype)
    return init


def constant_initializer(value=0.0, dtype=dtypes.float32):
    def init(shape, dtype_arg=None, partition_info=None):
        return ops.constant(value, dtype=dtype_arg or dtype, shape=shape)
    return init


def random_uniform_initializer(minval=0.0, maxval=1.0, seed=None,
                               dtype=dtypes.float32):
    def init(shape, dtype_arg=None, partition_info=None):
        return random_ops.random_uniform(shape, minval, maxval,
                                         dtype_arg or dtype, seed=seed)
    return init


def random_normal_initializer(mean=0.0, stddev=1.0, seed=None,
                              dtype=dtypes.float32):
    def init(shape, dtype_arg=None, partition_info=None):
        return random_ops.random_normal(shape, mean, stddev,
                                        dtype_arg or dtype, seed=seed)
    return init


def truncated_normal_initializer(mean=0.0, stddev=1.0, seed=None,
                                 dtype=dtypes.float32):
    def init(shape, dtype_arg=None, partition_info=None):
        return random_ops.truncated_normal(shape, mean, stddev,
                                           dtype_arg or dtype, seed=seed)
    return init


def _fans(shape):
    if len(shape) < 1:
        return 1, 1
    if len(shape) == 1:
        return shape[0], shape[0]
    if len(shape) == 2:
        return shape[0], shape[1]
    receptive = 1
    for d in shape[:-2]:
        receptive *= d
    return shape[-2] * receptive, shape[-1] * receptive


def glorot_uniform_initializer(seed=None, dtype=dtypes.float32):
    def init(shape, dtype_arg=None, partition_info=None):
        fan_in, fan_out = _fans(shape)
        limit = math.sqrt(6.0 / (fan_in + fan_out))
        return random_ops.random_uniform(shape, -limit, limit,
                                         dtype_arg or dtype, seed=seed)
    return init


xavier_initializer = glorot_uniform_initializer


def variance_scaling_initializer(scale=2.0, mode='fan_in', seed=None,
                                 dtype=dtypes.float32):
    def init(shape, dtype_arg=None, partition_info=None):
        fan_in, fan_out = _fans(shape)
        n = fan_in if mode == 'fan_in' else fan_out
        stddev = math.sqrt(scale / n)
        return random_ops.truncated_normal(shape, 0.0, stddev,
                                           dtype_arg or dtype, seed=seed)
    return init
