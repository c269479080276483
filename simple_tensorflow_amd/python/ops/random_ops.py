"""Random ops on the Philox engine (analog of python/ops/random_ops.py)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import apply_op, convert_to_tensor


def _seeds(seed):
    g = ops.get_default_graph()
    if seed is None:
        if g.seed is None:
            return 0, 0
        return g.seed, id(g) & 0x7FFFFFFF
    return g.seed or 87654321, seed


def random_uniform(shape, minval=0.0, maxval=1.0, dtype=dtypes.float32,
                   seed=None, name=None):
    s1, s2 = _seeds(seed)
    dt = dtypes.as_dtype(dtype)
    shape_t = convert_to_tensor(shape, dtype=dtypes.int32) \
        if not isinstance(shape, ops.Tensor) else shape
    if dt.is_integer:
        return apply_op('RandomUniformInt', shape_t,
                        convert_to_tensor(minval, dtype=dt),
                        convert_to_tensor(maxval, dtype=dt),
                        seed=s1, seed2=s2, name=name)
    u = apply_op('RandomUniform', shape_t, seed=s1, seed2=s2, dtype=dt,
                 name=name)
    if not isinstance(shape, ops.Tensor):
        u.set_shape(list(shape))
    if minval == 0.0 and maxval == 1.0:
        return u
    from simple_tensorflow_amd.python.ops import math_ops
    return math_ops.add(math_ops.multiply(u, float(maxval - minval)),
                        float(minval))


def random_normal(shape, mean=0.0, stddev=1.0, dtype=dtypes.float32,
                  seed=None, name=None):
    s1, s2 = _seeds(seed)
    dt = dtypes.as_dtype(dtype)
    shape_t = convert_to_tensor(shape, dtype=dtypes.int32) \
        if not isinstance(shape, ops.Tensor) else shape
    z = apply_op('RandomStandardNormal', shape_t, seed=s1, seed2=s2, dtype=dt,
                 name=name)
    if not isinstance(shape, ops.Tensor):
        z.set_shape(list(shape))
    if mean == 0.0 and stddev == 1.0:
        return z
    from simple_tensorflow_amd.python.ops import math_ops
    return math_ops.add(math_ops.multiply(z, float(stddev)), float(mean))


def truncated_normal(shape, mean=0.0, stddev=1.0, dtype=dtypes.float32,
                     seed=None, name=None):
    s1, s2 = _seeds(seed)
    dt = dtypes.as_dtype(dtype)
    shape_t = convert_to_tensor(shape, dtype=dtypes.int32) \
        if not isinstance(shape, ops.Tensor) else shape
    z = apply_op('TruncatedNormal', shape_t, seed=s1, seed2=s2, dtype=dt,
                 name=name)
    if not isinstance(shape, ops.Tensor):
        z.set_shape(list(shape))
    if mean == 0.0 and stddev == 1.0:
        return z
    from simple_tensorflow_amd.python.ops import math_ops
    return math_ops.add(math_ops.multiply(z, float(stddev)), float(mean))


def set_random_seed(seed):
    ops.get_default_graph().seed = seed
