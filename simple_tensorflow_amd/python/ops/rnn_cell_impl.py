"""RNN cells + static unroll (analog of reference python/ops/rnn.py +
rnn_cell_impl.py — the reference trim only ships the _RNNCell ABC since cells
lived in contrib; these are the standard TF-1.0-era cell definitions,
bf16-compute/f32-master capable)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.ops import (array_ops, init_ops, math_ops,
                                              nn_ops, variable_scope)


class RNNCell(object):
    def _cast_cached(self, var, dtype):
        """Cast a master-f32 variable to the compute dtype ONCE per graph.

        Cells are called once per unrolled timestep; an uncached cast would
        re-materialize the full weight matrix every step (35 casts of an
        18M-element kernel per PTB step before this)."""
        cache = getattr(self, '_weight_cast_cache', None)
        if cache is None:
            cache = self._weight_cast_cache = {}
        key = (id(ops.get_default_graph()), var.name, dtype)
        t = cache.get(key)
        if t is None:
            t = math_ops.cast(var.ref(), dtype)
            cache[key] = t
        return t

    @property
    def state_size(self):
        raise NotImplementedError

    @property
    def output_size(self):
        raise NotImplementedError

    def zero_state(self, batch_size, dtype):
        raise NotImplementedError

    def __call__(self, inputs, state, scope=None):
        raise NotImplementedError


class BasicRNNCell(RNNCell):
    def __init__(self, num_units, activation=math_ops.tanh):
        self._num_units = num_units
        self._activation = activation

    @property
    def state_size(self):
        return self._num_units

    @property
    def output_size(self):
        return self._num_units

    def zero_state(self, batch_size, dtype):
        return array_ops.zeros([batch_size, self._num_units], dtype)

    def __call__(self, inputs, state, scope=None):
        with variable_scope.variable_scope(scope or 'basic_rnn_cell'):
            in_dim = inputs._shape[-1]
            w = variable_scope.get_variable(
                'kernel', [in_dim + self._num_units, self._num_units])
            b = variable_scope.get_variable(
                'bias', [self._num_units],
                initializer=init_ops.zeros_initializer())
            cat = array_ops.concat([inputs, state], 1)
            wk = w.ref()
            bk = b.ref()
            if inputs.dtype == dtypes.bfloat16:
                wk = math_ops.cast(wk, dtypes.bfloat16)
                bk = math_ops.cast(bk, dtypes.bfloat16)
            out = self._activation(
                nn_ops.bias_add(math_ops.matmul(cat, wk), bk))
            return out, out


class BasicLSTMCell(RNNCell):
    def __init__(self, num_units, forget_bias=1.0, activation=math_ops.tanh):
        self._num_units = num_units
        self._forget_bias = forget_bias
        self._activation = activation

    @property
    def state_size(self):
        return (self._num_units, self._num_units)

    @property
    def output_size(self):
        return self._num_units

    def zero_state(self, batch_size, dtype):
        return (array_ops.zeros([batch_size, self._num_units], dtype),
                array_ops.zeros([batch_size, self._num_units], dtype))

    def __call__(self, inputs, state, scope=None):
        c, h = state
        with variable_scope.variable_scope(scope or 'basic_lstm_cell'):
            in_dim = inputs._shape[-1]
            w = variable_scope.get_variable(
                'kernel', [in_dim + self._num_units, 4 * self._num_units])
            b = variable_scope.get_variable(
                'bias', [4 * self._num_units],
                initializer=init_ops.zeros_initializer())
            cat = array_ops.concat([inputs, h], 1)
            wk = w.ref()
            bk = b.ref()
            if inputs.dtype == dtypes.bfloat16:
                wk = self._cast_cached(w, dtypes.bfloat16)
                bk = self._cast_cached(b, dtypes.bfloat16)
            gates = nn_ops.bias_add(math_ops.matmul(cat, wk), bk)
            i, j, f, o = array_ops.split(gates, 4, axis=1)
            fb = ops.constant(self._forget_bias, dtype=gates.dtype)
            new_c = c * math_ops.sigmoid(f + fb) + \
                math_ops.sigmoid(i) * self._activation(j)
            new_h = self._activation(new_c) * math_ops.sigmoid(o)
            return new_h, (new_c, new_h)


class LSTMBlockCell(RNNCell):
    """Fused LSTM cell: the per-timestep pointwise math runs as ONE LSTMGates
    op (HIP LstmGatesKernel) instead of the composed split + ~12 elementwise
    ops of BasicLSTMCell, and the backward pass is one LSTMGatesGrad emitting
    the packed [B, 4H] dgates the dW/dx GEMMs consume directly.

    Capability analog of the reference's contrib/rnn LSTMBlockCell
    (lstm_ops.cc); numerics match BasicLSTMCell(activation=tanh) exactly
    (tests/test_rnn_fused.py)."""

    def __init__(self, num_units, forget_bias=1.0):
        self._num_units = num_units
        self._forget_bias = forget_bias

    @property
    def state_size(self):
        return (self._num_units, self._num_units)

    @property
    def output_size(self):
        return self._num_units

    def zero_state(self, batch_size, dtype):
        return (array_ops.zeros([batch_size, self._num_units], dtype),
                array_ops.zeros([batch_size, self._num_units], dtype))

    def __call__(self, inputs, state, scope=None):
        c, h = state
        with variable_scope.variable_scope(scope or 'lstm_block_cell'):
            in_dim = inputs._shape[-1]
            w = variable_scope.get_variable(
                'kernel', [in_dim + self._num_units, 4 * self._num_units])
            b = variable_scope.get_variable(
                'bias', [4 * self._num_units],
                initializer=init_ops.zeros_initializer())
            wk = w.ref()
            bk = b.ref()
            if inputs.dtype == dtypes.bfloat16:
                wk = self._cast_cached(w, dtypes.bfloat16)
                bk = self._cast_cached(b, dtypes.bfloat16)
            cat = array_ops.concat([inputs, h], 1)
            gates = nn_ops.bias_add(math_ops.matmul(cat, wk), bk)
            outs = ops.apply_op('LSTMGates', gates, c,
                                forget_bias=float(self._forget_bias))
            i, f, o, ci, cs, co, new_h = outs
            for t in outs:
                t.set_shape(c._shape)
            return new_h, (cs, new_h)


class MultiRNNCell(RNNCell):
    def __init__(self, cells, state_is_tuple=True):
        self._cells = cells

    @property
    def output_size(self):
        return self._cells[-1].output_size

    def zero_state(self, batch_size, dtype):
        return tuple(c.zero_state(batch_size, dtype) for c in self._cells)

    def __call__(self, inputs, state, scope=None):
        new_states = []
        x = inputs
        with variable_scope.variable_scope(scope or 'multi_rnn_cell'):
            for i, (cell, st) in enumerate(zip(self._cells, state)):
                with variable_scope.variable_scope('cell_%d' % i):
                    x, ns = cell(x, st)
                new_states.append(ns)
        return x, tuple(new_states)


def static_rnn(cell, inputs, initial_state=None, dtype=None, scope=None):
    """Statically-unrolled RNN (the reference's tf.nn.rnn; the PTB config's
    execution style — while-loop dynamic_rnn is available for parity but the
    unrolled graph is what the MI355X scheduler pipelines best)."""
    outputs = []
    state = initial_state
    if state is None:
        batch = inputs[0]._shape[0]
        state = cell.zero_state(batch, dtype or inputs[0].dtype)
    g = ops.get_default_graph()
    with variable_scope.variable_scope(scope or 'rnn') as vs:
        for t, x in enumerate(inputs):
            if t == 1:
                vs.reuse = True
            out, state = cell(x, state)
            outputs.append(out)
    return outputs, state


def dynamic_rnn(cell, inputs, sequence_length=None, initial_state=None,
                dtype=None, time_major=False, scope=None,
                parallel_iterations=None, swap_memory=False):
    """while_loop + TensorArray RNN (reference python/ops/rnn.py
    dynamic_rnn; forward-only in round 1 — training uses static_rnn)."""
    from simple_tensorflow_amd.python.ops import control_flow_ops, math_ops
    from simple_tensorflow_amd.python.ops import tensor_array_ops
    from simple_tensorflow_amd.python.util import nest
    from simple_tensorflow_amd.python.framework import dtypes as _dt

    if not time_major:
        inputs = array_ops.transpose(inputs, [1, 0, 2])
    in_shape = inputs._shape
    T = in_shape[0] if in_shape is not None else None
    batch = in_shape[1] if in_shape is not None else None
    if T is None:
        raise ValueError('dynamic_rnn requires a static time dimension '
                         'in round 1')
    state = initial_state
    if state is None:
        state = cell.zero_state(batch, dtype or inputs.dtype)
    flat_state = nest.flatten(state)

    input_ta = tensor_array_ops.TensorArray(inputs.dtype, size=T)
    input_ta = input_ta.unstack(inputs)
    output_ta = tensor_array_ops.TensorArray(
        dtype or inputs.dtype, size=T)

    g = ops.get_default_graph()
    with variable_scope.variable_scope(scope or 'rnn') as vs:
        built = [False]

        def body(t, out_flow, *fs):
            st = nest.pack_sequence_as(state, list(fs))
            x = input_ta.read(t)
            if in_shape is not None:
                x.set_shape([in_shape[1], in_shape[2]])
            if built[0]:
                vs.reuse = True
            out, new_st = cell(x, st)
            built[0] = True
            ta = output_ta._with_flow(out_flow)
            new_flow = ta.write(t, out)._flow
            return [math_ops.add(t, 1), new_flow] + nest.flatten(new_st)

        results = control_flow_ops.while_loop(
            lambda t, f, *fs: math_ops.less(t, T), body,
            [ops.constant(0, dtype=_dt.int32), output_ta._flow] + flat_state)
    final_flow = results[1]
    final_state = nest.pack_sequence_as(state, list(results[2:]))
    outputs = output_ta._with_flow(final_flow).stack()
    osh = None
    if in_shape is not None and getattr(cell, 'output_size', None):
        outputs.set_shape([T, in_shape[1], cell.output_size])
    if not time_major:
        outputs = array_ops.transpose(outputs, [1, 0, 2])
    return outputs, final_state
