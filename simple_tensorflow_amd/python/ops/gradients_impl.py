"""Reverse-mode autodiff over the graph.

Analog of the reference's python/ops/gradients_impl.py (gradients:376):
reverse BFS from ys to xs, per-op grad functions from the registry, AddN
aggregation of fan-in gradients.
"""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.ops import array_ops, math_ops

# Import grad registrations for side effects.
from simple_tensorflow_amd.python.ops import math_grad  # noqa: F401
from simple_tensorflow_amd.python.ops import array_grad  # noqa: F401
from simple_tensorflow_amd.python.ops import nn_grad  # noqa: F401


def gradients(ys, xs, grad_ys=None, name='gradients',
              colocate_gradients_with_ops=False, gate_gradients=False,
              aggregation_method=None, stop_gradients=None):
    if not isinstance(ys, (list, tuple)):
        ys = [ys]
    single_x = not isinstance(xs, (list, tuple))
    if single_x:
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
        # ---- find ops between xs and ys ----
        x_ops = {x.op for x in xs}
        reached = set()  # ops from which some x is reachable (downstream sweep)
        stack = list(x_ops)
        consumers = _build_consumers(g)
        while stack:
            op = stack.pop()
            if op in reached:
                continue
            reached.add(op)
            for c in consumers.get(op, ()):  # ops consuming op's outputs
                stack.append(c)
        # ops that influence ys
        useful = set()
        stack = [y.op for y in ys]
        while stack:
            op = stack.pop()
            if op in useful:
                continue
            useful.add(op)
            for t in op.inputs:
                stack.append(t.op)
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
            for c in consumers.get(op, ()):  # consumer ops
                if c in between:
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

        while ready:
            op = ready.pop()
            if op in processed:
                continue
            processed.add(op)
            out_grads = []
            has_any = False
            for t in op.outputs:
                lst = grads.get(t)
                if lst:
                    has_any = True
                    out_grads.append(lst[0] if len(lst) == 1
                                     else math_ops.add_n(lst))
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
            for t, dg in zip(op.inputs, in_grads):
                if t in stop_set or t.op.type in ('StopGradient',):
                    dg = None
                if t.op in between:
                    add_grad(t, dg)
                    out_consumers[t.op] -= 1
                    if out_consumers[t.op] == 0:
                        ready.append(t.op)

        # ---- collect ----
        result = []
        for x in xs:
            lst = grads.get(x)
            if lst is None:
                # maybe gradient was recorded against the variable snapshot
                result.append(None)
            elif len(lst) == 1:
                result.append(lst[0])
            else:
                result.append(math_ops.add_n(lst))
        return result[0] if single_x else result


def _needs_fill(op):
    # ops whose grad fn requires every output grad slot present
    return op.type in ('Switch', 'Merge', 'SoftmaxCrossEntropyWithLogits',
                       'SparseSoftmaxCrossEntropyWithLogits', 'FusedBatchNorm')


def _build_consumers(g):
    consumers = {}
    for op in g._node_list:
        for t in dict.fromkeys(op.inputs):
            consumers.setdefault(t.op, []).append(op)
    return consumers
