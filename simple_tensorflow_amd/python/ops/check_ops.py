"""Assert/Print/check-numerics surface (reference python/ops/check_ops.py +
logging_ops.py): Assert and Print ride the py_func bridge (host-side side
effects, exactly where the reference runs them)."""
import numpy as np

from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import convert_to_tensor
from simple_tensorflow_amd.python.ops import math_ops, script_ops


def Assert(condition, data, summarize=3, name=None):
    from simple_tensorflow_amd.python.ops import control_flow_ops
    return control_flow_ops.Assert(condition, data, summarize=summarize,
                                   name=name)


def Print(input_, data, message=None, first_n=None, summarize=3, name=None):
    from simple_tensorflow_amd.python.framework.ops import apply_op
    t = convert_to_tensor(input_)
    out = apply_op('Print', t, [convert_to_tensor(d) for d in data],
                   message=message or '', first_n=first_n if first_n
                   is not None else -1, summarize=summarize, name=name)
    if t._shape is not None:
        out.set_shape(t._shape)
    return out


def _binary_assert(name, op_fn, x, y, data, summarize, msg):
    x = convert_to_tensor(x)
    y = convert_to_tensor(y, dtype=x.dtype)
    cond = math_ops.reduce_all(op_fn(x, y))
    if data is None:
        data = [x, y]
    return Assert(cond, data, summarize=summarize, name=name)


def assert_equal(x, y, data=None, summarize=3, message=None, name=None):
    return _binary_assert('assert_equal', math_ops.equal, x, y, data,
                          summarize, message)


def assert_less(x, y, data=None, summarize=3, message=None, name=None):
    return _binary_assert('assert_less', math_ops.less, x, y, data,
                          summarize, message)


def assert_greater(x, y, data=None, summarize=3, message=None, name=None):
    return _binary_assert('assert_greater', math_ops.greater, x, y, data,
                          summarize, message)


def assert_positive(x, data=None, summarize=3, message=None, name=None):
    x = convert_to_tensor(x)
    cond = math_ops.reduce_all(math_ops.greater(
        x, ops.constant(0, x.dtype)))
    return Assert(cond, data if data is not None else [x],
                  summarize=summarize, name=name or 'assert_positive')


def assert_non_negative(x, data=None, summarize=3, message=None, name=None):
    x = convert_to_tensor(x)
    cond = math_ops.reduce_all(math_ops.greater_equal(
        x, ops.constant(0, x.dtype)))
    return Assert(cond, data if data is not None else [x],
                  summarize=summarize, name=name or 'assert_non_negative')


def add_check_numerics_ops():
    """Attach a CheckNumerics to every float tensor in the graph (reference
    python/ops/numerics.py add_check_numerics_ops); returns a grouped op."""
    from simple_tensorflow_amd.python.framework.ops import apply_op
    from simple_tensorflow_amd.python.ops import control_flow_ops
    g = ops.get_default_graph()
    checks = []
    for op in list(g._node_list):
        if op.type in ('CheckNumerics', 'PyFunc'):
            continue
        for t in op.outputs:
            if t.dtype in (dtypes.float32, dtypes.float64, dtypes.bfloat16,
                           dtypes.float16):
                checks.append(apply_op(
                    'CheckNumerics', t,
                    message='%s:%d' % (op.name, t.value_index)).op)
    return control_flow_ops.group(*checks) if checks else None
