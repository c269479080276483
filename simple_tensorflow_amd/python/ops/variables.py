"""tf.Variable (analog of reference python/ops/variables.py Variable:33)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import convert_to_tensor
from simple_tensorflow_amd.python.ops import array_ops, state_ops


class Variable(object):
    def __init__(self, initial_value=None, trainable=True, collections=None,
                 validate_shape=True, name=None, dtype=None,
                 caching_device=None, expected_shape=None):
        g = ops.get_default_graph()
        with g.name_scope(name or 'Variable') as scope:
            if callable(initial_value):
                initial_value = initial_value()
            self._initial_value = convert_to_tensor(initial_value,
                                                    dtype=dtype,
                                                    name='initial_value')
            shape = self._initial_value._shape
            if shape is None or any(d is None for d in shape):
                raise ValueError('Variable needs fully-defined initial shape')
            self._variable = state_ops.variable_op(
                shape, self._initial_value.dtype,
                name=scope if scope else 'Variable')
            self._initializer_op = state_ops.assign(
                self._variable, self._initial_value,
                validate_shape=validate_shape).op
            self._snapshot = array_ops.identity(self._variable, name='read')
        cols = list(collections) if collections is not None else [
            ops.GraphKeys.GLOBAL_VARIABLES]
        if trainable and ops.GraphKeys.TRAINABLE_VARIABLES not in cols:
            cols.append(ops.GraphKeys.TRAINABLE_VARIABLES)
        for c in cols:
            g.add_to_collection(c, self)

    @classmethod
    def _from_graph_elements(cls, variable_tensor, initializer_op, snapshot):
        """Rebuild a Variable wrapper around already-imported graph nodes
        (used by meta_graph.import_meta_graph; analog of reference
        Variable.from_proto)."""
        v = cls.__new__(cls)
        v._variable = variable_tensor
        v._initializer_op = initializer_op
        v._snapshot = snapshot
        v._initial_value = None
        return v

    @property
    def name(self):
        return self._variable.name

    @property
    def dtype(self):
        return self._variable.dtype

    @property
    def op(self):
        return self._variable.op

    @property
    def graph(self):
        return self._variable.graph

    @property
    def shape(self):
        return self._variable.shape

    def get_shape(self):
        return self._variable.shape

    @property
    def initializer(self):
        return self._initializer_op

    @property
    def initial_value(self):
        return self._initial_value

    def value(self):
        return self._snapshot

    def read_value(self):
        return self._snapshot

    def ref(self):
        return self._variable

    def _as_graph_element(self):
        return self._variable

    def assign(self, value, use_locking=False):
        return state_ops.assign(self._variable, value)

    def assign_add(self, delta, use_locking=False):
        return state_ops.assign_add(self._variable, delta)

    def assign_sub(self, delta, use_locking=False):
        return state_ops.assign_sub(self._variable, delta)

    def eval(self, session=None):
        return self._snapshot.eval(session=session)

    def count_up_to(self, limit):
        from simple_tensorflow_amd.python.framework.ops import apply_op
        return apply_op('CountUpTo', self._variable, limit=limit)

    # math sugar
    def __add__(self, o): return self.value() + o
    def __radd__(self, o): return o + self.value()
    def __sub__(self, o): return self.value() - o
    def __rsub__(self, o): return o - self.value()
    def __mul__(self, o): return self.value() * o
    def __rmul__(self, o): return o * self.value()
    def __neg__(self): return -self.value()

    def __repr__(self):
        return "<Variable '%s' shape=%s>" % (self.name, self._variable._shape)


def global_variables(scope=None):
    return ops.get_default_graph().get_collection(
        ops.GraphKeys.GLOBAL_VARIABLES, scope)


all_variables = global_variables


def trainable_variables(scope=None):
    return ops.get_default_graph().get_collection(
        ops.GraphKeys.TRAINABLE_VARIABLES, scope)


def local_variables():
    return ops.get_default_graph().get_collection(
        ops.GraphKeys.LOCAL_VARIABLES)


def moving_average_variables():
    return ops.get_default_graph().get_collection(
        ops.GraphKeys.MOVING_AVERAGE_VARIABLES)


def variables_initializer(var_list, name='init'):
    from simple_tensorflow_amd.python.ops import control_flow_ops
    if not var_list:
        return control_flow_ops.no_op(name=name)
    return control_flow_ops.group(*[v.initializer for v in var_list],
                                  name=name)


def global_variables_initializer():
    return variables_initializer(global_variables())


initialize_all_variables = global_variables_initializer


def local_variables_initializer():
    return variables_initializer(local_variables())


def is_variable_initialized(variable):
    from simple_tensorflow_amd.python.ops import state_ops as so
    return so.is_variable_initialized(variable.ref())
