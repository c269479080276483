"""Reverse-mode autodiff over the graph.

Analog of the reference's python/ops/gradients_impl.py (gradients:376):
reverse BFS from ys to xs, per-op grad functions from the registry, AddN
aggregation of fan-in gradients.
"""
import os

from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.ops import array_ops, math_ops

# Import grad registrations for side effects.
from simple_tensorflow_amd.python.ops import math_grad  # noqa: F401
from simple_tensorflow_amd.python.ops import array_grad  # noqa: F401
from simple_tensorflow_amd.python.ops import nn_grad  # noqa: F401


class _PseudoOp(object):
    """Base for reverse-sweep stand-ins that collapse a subgraph (a while
    loop, a Defun call) into one differentiable node."""


class _WhilePseudoOp(_PseudoOp):
    """Stands in for a whole while loop during the reverse sweep: inputs are
    the original loop variables + captured externals, outputs the exits
    (reference gradients_impl + control_flow_grad collapse into this since
    our while_loop records its iteration inputs in TensorArrays)."""

    def __init__(self, record):
        self.record = record
        self.inputs = list(record['loop_var_inputs']) + \
            list(record['externals'])
        self.outputs = list(record['exits'])
        self.type = '_WhileLoop'
        self.name = record['exits'][0].op.name + '_loop'


class _DefunPseudoOp(_PseudoOp):
    """Stands in for one Defun call with a custom gradient: the reverse
    sweep treats the whole instantiated body as a single op and invokes
    grad_func / python_grad_func (the reference's SymbolicGradient node,
    python/framework/function.py)."""

    def __init__(self, record):
        self.record = record
        self.inputs = list(record.inputs)
        self.outputs = list(record.outputs)
        self.type = '_DefunCall'
        self.name = record.outputs[0].op.name + '_call'


def _while_grad(record, exit_grads):
    """Builds the backward loop: j = N-1..0, reading iteration inputs from
    the forward recording and rematerializing the body to differentiate it.
    Returns grads for loop_var_inputs + externals."""
    from simple_tensorflow_amd.python.ops import control_flow_ops
    from simple_tensorflow_amd.python.ops import tensor_array_ops
    n = record['n']
    ext = record['externals']
    body_fn = record['body_fn']
    count = record['count_exit']
    tas = [record['tas'][k]._with_flow(record['ta_flow_exits'][k])
           for k in range(n)]
    g = ops.get_default_graph()
    with g.name_scope(record['exits'][0].op.name.rsplit('/', 1)[0] +
                      '_grad'):
        gvars0 = [eg if eg is not None else
                  array_ops.zeros_like(record['exits'][k])
                  for k, eg in enumerate(exit_grads)]
        gext0 = [array_ops.zeros_like(e) for e in ext]

        def gcond(j, *rest):
            return math_ops.greater_equal(j, ops.constant(0))

        def gbody(j, *rest):
            gvars = list(rest[:n])
            gexts = list(rest[n:])
            vals = [tas[k].read(j) for k in range(n)]
            for k, v in enumerate(vals):
                if record['exits'][k]._shape is not None:
                    v.set_shape(record['exits'][k]._shape)
            from simple_tensorflow_amd.python.ops import variable_scope \
                as vs_mod
            with vs_mod.variable_scope(record['var_scope'], reuse=True):
                outs = body_fn(*vals)
            if not isinstance(outs, (list, tuple)):
                outs = [outs]
            outs = [ops.convert_to_tensor(o) for o in outs]
            inner = gradients(outs, list(vals) + list(ext), grad_ys=gvars,
                              name='bodygrad')
            new_gvars = [ig if ig is not None else array_ops.zeros_like(v)
                         for ig, v in zip(inner[:n], vals)]
            new_gexts = [ge + ig if ig is not None else ge
                         for ge, ig in zip(gexts, inner[n:])]
            return [math_ops.subtract(j, 1)] + new_gvars + new_gexts

        res = control_flow_ops.while_loop(
            gcond, gbody,
            [count - ops.constant(1)] + gvars0 + gext0,
            back_prop=False)
        if not isinstance(res, (list, tuple)):
            res = [res]
        return list(res[1:1 + n]) + list(res[1 + n:])


def _aggregate(lst):
    """Sum a fan-in gradient list. Two-term sums where one side is a
    freshly built Conv2DBackpropInput (the ResNet residual-block pattern:
    dx_branch + d_shortcut) are folded into Conv2DBackpropInputAdd so the
    add runs in the conv GEMM's epilogue instead of a separate
    full-tensor pass."""
    if len(lst) == 1:
        return lst[0]
    if len(lst) == 2 and not os.environ.get('STF_NO_CONV_DX_FUSE'):
        fused = (_fuse_conv_dx_add(lst[0], lst[1]) or
                 _fuse_conv_dx_add(lst[1], lst[0]))
        if fused is not None:
            return fused
    return math_ops.add_n(lst)


def _fuse_conv_dx_add(conv_t, side):
    op = getattr(conv_t, 'op', None)
    if op is None or getattr(op, 'type', None) != 'Conv2DBackpropInput':
        return None
    if conv_t is side or conv_t.dtype != side.dtype:
        return None
    cs, ss = conv_t._shape, side._shape
    if (cs is None or ss is None or any(d is None for d in cs) or
            list(cs) != list(ss)):
        return None
    # only when this partial gradient has no other use: otherwise the
    # rewrite would duplicate the conv GEMM
    if len(conv_t.consumers()) != 0:
        return None
    out = ops.apply_op(
        'Conv2DBackpropInputAdd', op.inputs[0], op.inputs[1], op.inputs[2],
        side, strides=op.get_attr('strides'), padding=op.get_attr('padding'))
    out.set_shape(cs)
    return out


def gradients(ys, xs, grad_ys=None, name='gradients',
              colocate_gradients_with_ops=False, gate_gradients=False,
              aggregation_method=None, stop_gradients=None):
    if not isinstance(ys, (list, tuple)):
        ys = [ys]
    # Reference contract (python/ops/gradients_impl.py gradients:376): the
    # result is ALWAYS a list, even for a single non-list x — unwrapping the
    # single case makes tf.gradients(loss, w)[0] silently index into the
    # gradient tensor.
    if not isinstance(xs, (list, tuple)):
        xs = [xs]
    xs = [x._as_graph_element() if hasattr(x, '_as_graph_element') else x
          for x in xs]
    ys = [ops.convert_to_tensor(y) for y in ys]
    if grad_ys is None:
        grad_ys = [None] * len(ys)
    elif not isinstance(grad_ys, (list, tuple)):
        grad_ys = [grad_ys]
    stop_set = set()
    if stop_gradients:
        for s in stop_gradients:
            stop_set.add(s)

    g = ops.get_default_graph()
    with g.name_scope(name):
        # ---- collapse recorded while loops into pseudo ops ----
        pseudo_of = {}   # internal/exit op -> _WhilePseudoOp
        pseudos = []
        for op in g._node_list:
            if op.type != 'Exit':
                continue
            rec = getattr(op.outputs[0], '_while_record', None) if \
                op.outputs else None
            if rec is None or id(rec) in {id(p.record) for p in pseudos}:
                continue
            ps = _WhilePseudoOp(rec)
            pseudos.append(ps)
            for iop in rec['internal_ops']:
                pseudo_of[iop] = ps

        # ---- collapse Defun calls with custom gradients ----
        seen_defun = set()
        for op in g._node_list:
            for t in op.outputs:
                rec = getattr(t, '_defun_record', None)
                if rec is None or id(rec) in seen_defun:
                    continue
                seen_defun.add(id(rec))
                ps = _DefunPseudoOp(rec)
                pseudos.append(ps)
                for iop in rec.internal_ops:
                    # while-loop collapse wins if an op is inside both
                    pseudo_of.setdefault(iop, ps)

        def xlate(op):
            return pseudo_of.get(op, op)

        # ---- find ops between xs and ys (pseudo-aware) ----
        x_ops = {xlate(x.op) for x in xs}
        consumers = _build_consumers(g)

        def successors(node):
            if isinstance(node, _PseudoOp):
                outs = node.outputs
                seen = []
                for t in outs:
                    for c in consumers.get(t.op, ()):  # consumers of exits
                        seen.append(xlate(c))
                return seen
            return [xlate(c) for c in consumers.get(node, ())]

        def predecessors(node):
            ins = node.inputs
            return [xlate(t.op) for t in ins]

        reached = set()
        stack = list(x_ops)
        while stack:
            op = stack.pop()
            if op in reached:
                continue
            reached.add(op)
            stack.extend(successors(op))
        useful = set()
        stack = [xlate(y.op) for y in ys]
        while stack:
            op = stack.pop()
            if op in useful:
                continue
            useful.add(op)
            stack.extend(predecessors(op))
        between = reached & useful

        # ---- init output grads ----
        grads = {}  # Tensor -> list of grad tensors to sum

        def add_grad(t, dg):
            if dg is None:
                return
            grads.setdefault(t, []).append(dg)

        for y, gy in zip(ys, grad_ys):
            if gy is None:
                gy = array_ops.ones_like(y)
            else:
                gy = ops.convert_to_tensor(gy)
            add_grad(y, gy)

        # ---- pending counts: number of relevant consumers per op output ----
        pending = {}
        for op in between:
            cnt = 0
            for t in op.inputs:
                if t.op in between:
                    pass
            # count consumers in `between` for this op's outputs
        # (we count per-op: how many times its outputs are consumed by ops in
        #  `between` that will eventually hand a gradient back)
        out_consumers = {}
        for op in between:
            n = 0
            outs = op.outputs
            for t_out in outs:
                for c in consumers.get(
                        t_out.op if isinstance(op, _PseudoOp) else op,
                        ()):
                    if xlate(c) in between and xlate(c) is not op:
                        n += sum(1 for t in c.inputs if t is t_out)
                if not isinstance(op, _PseudoOp):
                    break  # plain op: counted all outputs via consumers map
            if not isinstance(op, _PseudoOp):
                n = 0
                for c in consumers.get(op, ()):
                    if xlate(c) in between:
                        n += sum(1 for t in c.inputs if t.op is op)
            out_consumers[op] = n

        # seed: ops whose outputs include ys get their seed grads immediately
        ready = []
        processed = set()
        y_ops = {y.op for y in ys}

        def op_ready(op):
            # an op is ready when all its relevant consumers have contributed
            return out_consumers.get(op, 0) <= 0

        for yop in y_ops:
            if yop in between:
                # consumers of y that are in between still pending; but the y
                # seed itself doesn't wait
                pass
        # Kahn-style: start from ops with zero relevant consumers
        for op in between:
            if out_consumers[op] == 0:
                ready.append(op)

        def _propagate(op, in_grads):
            # Hand a (possibly None) gradient to every input and decrement
            # the producer's consumer count — Kahn readiness depends on the
            # decrement happening even when the grad is None.
            for t, dg in zip(op.inputs, in_grads):
                if dg is not None and (
                        t in stop_set or t.op.type in ('StopGradient',)):
                    dg = None
                tprod = xlate(t.op)
                if tprod in between:
                    add_grad(t, dg)
                    out_consumers[tprod] -= 1
                    if out_consumers[tprod] == 0:
                        ready.append(tprod)

        while ready:
            op = ready.pop()
            if op in processed:
                continue
            processed.add(op)
            if isinstance(op, _DefunPseudoOp):
                rec = op.record
                out_grads = []
                for t in op.outputs:
                    lst = grads.get(t)
                    out_grads.append(None if not lst
                                     else _aggregate(lst))
                if all(og is None for og in out_grads):
                    # Call not on any differentiated path: still propagate
                    # None to inputs so upstream producers become ready.
                    _propagate(op, [None] * len(op.inputs))
                    continue
                filled = [og if og is not None else array_ops.zeros_like(t)
                          for t, og in zip(op.outputs, out_grads)]
                with g.name_scope(op.name + '_grad'):
                    if rec.func.python_grad_func is not None:
                        res = rec.func.python_grad_func(rec, *filled)
                    else:
                        res = rec.func.grad_func(*(list(op.inputs) + filled))
                in_grads = list(res) if isinstance(res, (list, tuple)) \
                    else [res]
                if len(in_grads) != len(op.inputs):
                    raise ValueError(
                        'gradient of function %s returned %d grads, want %d'
                        % (rec.func.name, len(in_grads), len(op.inputs)))
                _propagate(op, in_grads)
                continue
            if isinstance(op, _WhilePseudoOp):
                exit_grads = []
                for t in op.outputs:
                    lst = grads.get(t)
                    exit_grads.append(None if not lst
                                      else _aggregate(lst))
                if all(eg is None for eg in exit_grads):
                    # Loop not on any differentiated path: still propagate
                    # None so upstream producers become ready.
                    _propagate(op, [None] * len(op.inputs))
                    continue
                in_grads = _while_grad(op.record, exit_grads)
                _propagate(op, in_grads)
                continue
            out_grads = []
            has_any = False
            for t in op.outputs:
                lst = grads.get(t)
                if lst:
                    has_any = True
                    out_grads.append(_aggregate(lst))
                else:
                    out_grads.append(None)
            in_grads = [None] * len(op.inputs)
            if has_any and op.inputs:
                grad_fn = ops.get_gradient_function(op.type)
                if grad_fn == '__missing__':
                    raise LookupError('No gradient defined for op %s (%s)'
                                      % (op.type, op.name))
                if grad_fn is not None:
                    # fill missing output grads with zeros for multi-output ops
                    filled = []
                    for t, gt in zip(op.outputs, out_grads):
                        if gt is None and _needs_fill(op):
                            filled.append(array_ops.zeros_like(t))
                        else:
                            filled.append(gt)
                    with g.name_scope(op.name + '_grad'):
                        res = grad_fn(op, *filled)
                    if not isinstance(res, (list, tuple)):
                        res = [res]
                    in_grads = list(res)
                    if len(in_grads) != len(op.inputs):
                        raise ValueError(
                            'grad fn for %s returned %d grads, want %d' %
                            (op.type, len(in_grads), len(op.inputs)))
            _propagate(op, in_grads)

        # ---- collect ----
        result = []
        for x in xs:
            lst = grads.get(x)
            if lst is None:
                # maybe gradient was recorded against the variable snapshot
                result.append(None)
            else:
                result.append(_aggregate(lst))
        return result


def _needs_fill(op):
    # ops whose grad fn requires every output grad slot present
    return op.type in ('Switch', 'Merge', 'SoftmaxCrossEntropyWithLogits',
                       'SparseSoftmaxCrossEntropyWithLogits', 'FusedBatchNorm',
                       'LSTMGates')


def _build_consumers(g):
    consumers = {}
    for op in g._node_list:
        for t in dict.fromkeys(op.inputs):
            consumers.setdefault(t.op, []).append(op)
    return consumers
