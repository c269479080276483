"""State ops (analog of python/ops/state_ops.py)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import apply_op, convert_to_tensor


def variable_op(shape, dtype, name='Variable', container='', shared_name=''):
    t = apply_op('VariableV2', shape=list(shape), dtype=dtype, name=name,
                 container=container, shared_name=shared_name)
    t.set_shape(shape)
    return t


def assign(ref, value, validate_shape=True, use_locking=True, name=None):
    if hasattr(ref, '_variable'):  # tf.Variable → its mutable ref tensor
        ref = ref._variable
    t = apply_op('Assign', ref, convert_to_tensor(value, dtype=ref.dtype),
                 validate_shape=validate_shape, use_locking=use_locking,
                 name=name)
    t.set_shape(ref._shape)
    return t


def assign_add(ref, value, use_locking=False, name=None):
    return apply_op('AssignAdd', ref, convert_to_tensor(value, dtype=ref.dtype),
                    use_locking=use_locking, name=name)


def assign_sub(ref, value, use_locking=False, name=None):
    return apply_op('AssignSub', ref, convert_to_tensor(value, dtype=ref.dtype),
                    use_locking=use_locking, name=name)


def scatter_add(ref, indices, updates, use_locking=False, name=None):
    return apply_op('ScatterAdd', ref, indices, updates,
                    use_locking=use_locking, name=name)


def scatter_sub(ref, indices, updates, use_locking=False, name=None):
    return apply_op('ScatterSub', ref, indices, updates,
                    use_locking=use_locking, name=name)


def is_variable_initialized(ref, name=None):
    return apply_op('IsVariableInitialized', ref, name=name)
