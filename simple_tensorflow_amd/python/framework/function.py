"""Graph functions: `Defun` + symbolic-gradient overrides.

Capability analog of the reference's python/framework/function.py (Defun:
~line 700) and the SymbolicGradient op. The reference serializes each
function as a FunctionDef in the GraphDef library and instantiates it at
runtime through a function-call kernel; gradients of a call are taken by a
SymbolicGradient node that re-instantiates the body.

MI355X-native redesign: calls are instantiated INLINE at the call site
(one uniquified name scope per call). The executor and hipGraph capture see
plain ops — a call costs nothing at run time and whole-step capture works
through function boundaries, which a call-kernel indirection would break.
Custom gradients (`grad_func` / `python_grad_func`) are honored by
collapsing each call's ops into a pseudo-op during the reverse sweep
(the same machinery that collapses while loops), standing in for the
reference's SymbolicGradient node.
"""
import functools

from simple_tensorflow_amd.python.framework import ops


class _DefunCallRecord(object):
    """One instantiated call: what `gradients()` needs to treat the whole
    call as a single differentiable op with a custom gradient."""

    def __init__(self, func, inputs, outputs, internal_ops):
        self.func = func
        self.inputs = list(inputs)
        self.outputs = list(outputs)
        self.internal_ops = internal_ops


class _DefinedFunction(object):
    """A graph function produced by @Defun. Calling it instantiates the
    python body inline under a fresh `<name>` scope."""

    def __init__(self, func, input_types, func_name=None, grad_func=None,
                 python_grad_func=None):
        self._func = func
        self._input_types = list(input_types)
        self.name = func_name or func.__name__
        self.grad_func = grad_func            # a _DefinedFunction
        self.python_grad_func = python_grad_func  # callable(call, *dys)
        functools.update_wrapper(self, func)

    def __call__(self, *args):
        if len(args) != len(self._input_types):
            raise ValueError('%s expects %d arguments, got %d' %
                             (self.name, len(self._input_types), len(args)))
        g = ops.get_default_graph()
        g._add_function(self)  # serialized into the GraphDef library
        tensors = [ops.convert_to_tensor(a, dtype=dt)
                   for a, dt in zip(args, self._input_types)]
        mark = len(g._node_list)
        with g.name_scope(self.name):
            outs = self._func(*tensors)
        single = not isinstance(outs, (list, tuple))
        outs = [outs] if single else list(outs)
        outs = [ops.convert_to_tensor(o) for o in outs]
        if self.grad_func is not None or self.python_grad_func is not None:
            # Collapse this call in the reverse sweep. Ops with no inputs
            # (constants, variables created inside) are excluded, mirroring
            # the while-loop records: hiding them would orphan any outside
            # consumers of the same nodes.
            internal = [op for op in g._node_list[mark:] if op.inputs]
            rec = _DefunCallRecord(self, tensors, outs, internal)
            for t in outs:
                t._defun_record = rec
        return outs[0] if single else tuple(outs)


    @property
    def definition(self):
        """Serialized FunctionDef (reference function.proto wire format):
        the body is traced into a private graph with `_arg_i` placeholders;
        ret maps `out_i` to the producing tensor names."""
        from simple_tensorflow_amd.python.framework import dtypes, pbwire
        g = ops.Graph()
        with g.as_default():
            from simple_tensorflow_amd.python.ops import array_ops
            args = [array_ops.placeholder(dt, name='_arg_%d' % i)
                    for i, dt in enumerate(self._input_types)]
            outs = self._func(*args)
        single = not isinstance(outs, (list, tuple))
        outs = [outs] if single else list(outs)
        node_bytes = [op.node_def_bytes() for op in g._node_list
                      if not op.name.startswith('_arg_')]
        in_args = [('_arg_%d' % i, dt.as_datatype_enum)
                   for i, dt in enumerate(self._input_types)]
        out_args = [('out_%d' % i, o.dtype.as_datatype_enum)
                    for i, o in enumerate(outs)]
        sig = pbwire.op_def_signature(self.name, in_args, out_args)
        ret = {'out_%d' % i: o.name for i, o in enumerate(outs)}
        return pbwire.function_def(sig, node_bytes, ret)


def from_function_def(fdef_bytes):
    """Parses a wire-format FunctionDef back into a callable graph function
    (instantiated inline at each call, like @Defun)."""
    from simple_tensorflow_amd.python.framework import dtypes, pbreader
    fdef = pbreader.parse_function_def(bytes(fdef_bytes))
    sig = fdef['signature']
    input_types = [dtypes.as_dtype(a['type']) for a in sig['input_arg']]
    n_out = len(sig['output_arg'])
    ret_names = [fdef['ret']['out_%d' % i] for i in range(n_out)]
    nodes = fdef['node_def']

    def body(*args):
        from simple_tensorflow_amd.python.framework import importer
        input_map = {'_arg_%d' % i: a for i, a in enumerate(args)}
        outs = importer.import_graph_def(nodes, input_map=input_map,
                                         return_elements=ret_names, name='')
        return outs[0] if n_out == 1 else tuple(outs)

    return _DefinedFunction(body, input_types, func_name=sig['name'])



class Defun(object):
    """Decorator: `@Defun(tf.float32, tf.float32)` turns a python function
    over tensors into a graph function.

        @function.Defun(tf.float32)
        def f(x):
            return x * x

        y = f(tf.constant(3.0))          # instantiates f inline
        dx, = tf.gradients(y, [x])       # differentiates through the body

    `grad_func`: a _DefinedFunction g(x1..xn, dy1..dym) -> (dx1..dxn) used
    instead of differentiating the body (the reference's SymbolicGradient).
    `python_grad_func`: callable(call_record, *dys) -> dxs, analog of a
    python-registered gradient.
    """

    def __init__(self, *input_types, **kwargs):
        self._input_types = input_types
        self._func_name = kwargs.pop('func_name', None)
        self._grad_func = kwargs.pop('grad_func', None)
        self._python_grad_func = kwargs.pop('python_grad_func', None)
        if kwargs:
            raise TypeError('unknown Defun arguments: %r' % sorted(kwargs))

    def __call__(self, func):
        return _DefinedFunction(func, self._input_types,
                                func_name=self._func_name,
                                grad_func=self._grad_func,
                                python_grad_func=self._python_grad_func)
