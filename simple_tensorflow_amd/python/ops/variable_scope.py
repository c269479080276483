"""tf.variable_scope / tf.get_variable (analog of reference
python/ops/variable_scope.py get_variable:900): named variable reuse."""
import contextlib

from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.ops import init_ops, variables


class _VarStore(object):
    def __init__(self):
        self.vars = {}


def _store():
    g = ops.get_default_graph()
    if not hasattr(g, '_var_store'):
        g._var_store = _VarStore()
    return g._var_store


class VariableScope(object):
    def __init__(self, name, reuse=False, initializer=None):
        self.name = name
        self.reuse = reuse
        self.initializer = initializer


_scope_stack = []


def get_variable_scope():
    if not _scope_stack:
        _scope_stack.append(VariableScope(''))
    return _scope_stack[-1]


@contextlib.contextmanager
def variable_scope(name_or_scope, default_name=None, reuse=None,
                   initializer=None, dtype=None):
    parent = get_variable_scope()
    if isinstance(name_or_scope, VariableScope):
        scope = VariableScope(name_or_scope.name,
                              reuse if reuse is not None
                              else name_or_scope.reuse,
                              initializer or name_or_scope.initializer)
    else:
        name = name_or_scope if name_or_scope is not None else default_name
        full = parent.name + '/' + name if parent.name else name
        scope = VariableScope(full,
                              reuse if reuse is not None else parent.reuse,
                              initializer or parent.initializer)
    _scope_stack.append(scope)
    g = ops.get_default_graph()
    try:
        with g.name_scope(scope.name + '/' if scope.name else ''):
            yield scope
    finally:
        _scope_stack.pop()


def get_variable(name, shape=None, dtype=dtypes.float32, initializer=None,
                 regularizer=None, trainable=True, collections=None,
                 partitioner=None, **kw):
    scope = get_variable_scope()
    full_name = scope.name + '/' + name if scope.name else name
    store = _store()
    if full_name in store.vars:
        if not scope.reuse:
            raise ValueError('Variable %s already exists (set reuse=True)'
                             % full_name)
        return store.vars[full_name]
    if scope.reuse:
        raise ValueError('Variable %s does not exist (reuse=True)' % full_name)
    if initializer is None:
        initializer = scope.initializer or \
            init_ops.glorot_uniform_initializer()
    if callable(initializer):
        init_val = initializer(shape, dtype)
    else:
        init_val = initializer
    v = variables.Variable(init_val, trainable=trainable,
                           collections=collections, name=full_name,
                           dtype=dtype if not callable(initializer) else None)
    store.vars[full_name] = v
    if regularizer is not None:
        loss = regularizer(v.value())
        if loss is not None:
            ops.get_default_graph().add_to_collection(
                ops.GraphKeys.REGULARIZATION_LOSSES, loss)
    return v
