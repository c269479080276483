"""Control flow: group, cond, while_loop (Switch/Merge/Enter/Exit/
NextIteration graphs — analog of reference python/ops/control_flow_ops.py
cond:1673, while_loop:2495, compacted to the dataflow-executor semantics in
csrc/runtime/executor.cc)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import apply_op, convert_to_tensor
from simple_tensorflow_amd.python.ops import array_ops, math_ops


def no_op(name=None):
    g = ops.get_default_graph()
    return g.create_op('NoOp', [], [], name=name or 'NoOp')


def group(*inputs, **kwargs):
    name = kwargs.pop('name', None)
    g = ops.get_default_graph()
    deps = []
    for x in inputs:
        if x is None:
            continue
        deps.append(x.op if isinstance(x, ops.Tensor) else x)
    return g.create_op('NoOp', [], [], name=name or 'group_deps',
                       control_inputs=deps)


def with_dependencies(dependencies, output_tensor, name=None):
    with ops.control_dependencies(dependencies):
        return array_ops.identity(output_tensor, name=name)


tuple_ = None  # tf.tuple is rarely used; omitted in round 1


def switch(data, pred, name=None):
    data = convert_to_tensor(data)
    res = apply_op('Switch', data, pred, name=name)
    for r in res:
        r.set_shape(data._shape)
    return res  # (output_false, output_true)


def merge(inputs, name=None):
    res = apply_op('Merge', [convert_to_tensor(v) for v in inputs], name=name)
    res[0].set_shape(inputs[0]._shape)
    return res  # (output, value_index)


def _enter(data, frame_name, is_constant=False, name=None):
    t = apply_op('Enter', data, frame_name=frame_name,
                 is_constant=is_constant, name=name)
    t.set_shape(data._shape)
    return t


def _exit(data, name=None):
    t = apply_op('Exit', data, name=name)
    t.set_shape(data._shape)
    return t


def _next_iteration(data, name=None):
    t = apply_op('NextIteration', data, name=name)
    t.set_shape(data._shape)
    return t


def loop_cond(pred, name=None):
    return apply_op('LoopCond', pred, name=name)


def cond(pred, fn1, fn2, name=None):
    """tf.cond: evaluates fn1() when pred is true, else fn2()."""
    g = ops.get_default_graph()
    with g.name_scope(name or 'cond'):
        pred = convert_to_tensor(pred)
        # Build both branches, gated by Switch on every external tensor use.
        # Compact approach: wrap branch outputs through Switch on pred, then
        # Merge. Side effects inside the untaken branch are still gated
        # because their results flow through the dead Switch port.
        p_f, p_t = switch(pred, pred)
        with g.name_scope('then'):
            res_t = fn1()
        with g.name_scope('else'):
            res_f = fn2()
        single = not isinstance(res_t, (list, tuple))
        if single:
            res_t, res_f = [res_t], [res_f]
        outs = []
        for t_val, f_val in zip(res_t, res_f):
            t_val = convert_to_tensor(t_val)
            f_val = convert_to_tensor(f_val, dtype=t_val.dtype)
            # gate each branch value by its switch port
            gated_t = switch(t_val, pred)[1]
            gated_f = switch(f_val, pred)[0]
            m, _ = merge([gated_f, gated_t])
            outs.append(m)
        return outs[0] if single else outs


class _WhileCtx(object):
    """Capture context: tensors from outside the frame are routed through
    constant Enter nodes (analog of the reference WhileContext.AddValue)."""

    def __init__(self, graph, frame):
        self.graph = graph
        self.frame = frame
        self.internal = set()
        self.emap = {}

    def capture(self, t):
        if t in self.internal:
            return t
        if t in self.emap:
            return self.emap[t]
        # A variable snapshot ('read' Identity of a ref) may itself live in
        # another frame; capture the underlying ref instead so gradients of
        # rematerialized bodies reach the variable (gradients_impl._while_grad).
        if t.op.type == 'Identity' and t.op.inputs:
            src = t.op.inputs[0]
            while src.op.type == 'Enter' and src.op.inputs:
                src = src.op.inputs[0]
            if getattr(src, '_is_ref', False):
                return self.capture(src)
        # build the Enter OUTSIDE the context to avoid recursion
        self.graph._while_ctx_stack.pop()
        try:
            e = _enter(t, self.frame, is_constant=True,
                       name=t.op.name.split('/')[-1] + '_enter')
        finally:
            self.graph._while_ctx_stack.append(self)
        cv = getattr(t, '_const_value', None)
        if cv is not None:
            e._const_value = cv  # keep static-shape grads working
        self.internal.add(e)
        self.emap[t] = e
        return e

    def mark(self, *tensors):
        for t in tensors:
            self.internal.add(t)


def while_loop(cond_fn, body_fn, loop_vars, shape_invariants=None,
               parallel_iterations=10, back_prop=True, swap_memory=False,
               name=None):
    """tf.while_loop with the standard Enter/Merge/Switch/Body/NextIteration
    ring (reference control_flow_ops.while_loop:2495 semantics on the
    executor's frame/iteration machinery).

    With back_prop=True the loop additionally carries an iteration counter
    and per-variable TensorArrays recording each iteration's inputs; the
    gradient pass (gradients_impl._while_grad) runs a second loop backwards
    over the recording, rematerializing the body at the saved values.
    """
    single = not isinstance(loop_vars, (list, tuple))
    if single:
        loop_vars = [loop_vars]
    user_n = len(loop_vars)
    record = None
    if back_prop:
        from simple_tensorflow_amd.python.ops import tensor_array_ops
        user_cond, user_body = cond_fn, body_fn
        loop_vars = [convert_to_tensor(v) for v in loop_vars]
        tas = [tensor_array_ops.TensorArray(v.dtype, size=0,
                                            dynamic_size=True)
               for v in loop_vars]
        aug = 1 + user_n  # [counter] + [ta flows]

        def cond_fn(i, *rest):  # noqa: F811
            return user_cond(*rest[user_n:])

        def body_fn(i, *rest):  # noqa: F811
            flows = rest[:user_n]
            vals = rest[user_n:]
            new_flows = [tas[k]._with_flow(flows[k]).write(i, vals[k]).flow
                         for k in range(user_n)]
            outs = user_body(*vals)
            if not isinstance(outs, (list, tuple)):
                outs = [outs]
            return [math_ops.add(i, 1)] + new_flows + list(outs)

        loop_vars = ([ops.constant(0, dtype=dtypes.int32)] +
                     [ta.flow for ta in tas] + loop_vars)

    g = ops.get_default_graph()
    with g.name_scope(name or 'while') as scope:
        frame = scope[:-1] if scope.endswith('/') else (scope or 'while')
        loop_vars = [convert_to_tensor(v) for v in loop_vars]
        enters = [_enter(v, frame) for v in loop_vars]
        ctx = _WhileCtx(g, frame)
        g._while_ctx_stack.append(ctx)
        n_before = len(g._node_list)
        try:
            merges = []
            for e in enters:
                m, idx = merge([e, e])  # input 1 rewired to NextIteration
                ctx.mark(m, idx)
                merges.append(m)
            p = cond_fn(*merges)
            p = loop_cond(p)
            ctx.mark(p)
            switches = []
            for m in merges:
                sw = switch(m, p)
                ctx.mark(*sw)
                switches.append(sw)
            body_in = [array_ops.identity(s[1]) for s in switches]
            for b in body_in:
                ctx.mark(b)
            body_out = body_fn(*body_in)
            if not isinstance(body_out, (list, tuple)):
                body_out = [body_out]
            body_out = [convert_to_tensor(v) for v in body_out]
            nexts = [_next_iteration(v) for v in body_out]
        finally:
            g._while_ctx_stack.pop()
        # Zero-input ops (Const/VariableV2) created during the body build
        # execute in the root frame and may be consumed outside the loop —
        # they are NOT loop-internal (the gradient sweep must see them as
        # ordinary producers, not as part of the loop pseudo-op).
        internal_ops = {op for op in g._node_list[n_before:]
                        if op.inputs} | {e.op for e in enters}
        # rewire each merge's second input to the NextIteration tensor
        for m, n in zip(merges, nexts):
            m.op.inputs[1] = n
            g._bump_version(m.op)
        exits = [_exit(s[0]) for s in switches]
        internal_ops |= {e.op for e in exits}

    if back_prop:
        from simple_tensorflow_amd.python.ops import variable_scope as vs_mod
        user_exits = exits[1 + user_n:]
        record = {
            'cond_fn': user_cond,
            'body_fn': user_body,
            'var_scope': vs_mod.get_variable_scope(),
            'n': user_n,
            'loop_var_inputs': loop_vars[1 + user_n:],
            'count_exit': exits[0],
            'ta_flow_exits': exits[1:1 + user_n],
            'tas': tas,
            'exits': user_exits,
            'exit_ops': {e.op for e in user_exits} |
                        {exits[0].op} | {e.op for e in exits[1:1 + user_n]},
            'externals': list(ctx.emap.keys()),
            'internal_ops': internal_ops,
        }
        for e in user_exits:
            e._while_record = record
        exits = user_exits
    return exits[0] if (single or len(exits) == 1) else exits


def Assert(condition, data, summarize=3, name=None):  # noqa: N802
    data = [convert_to_tensor(d) for d in data]
    g = ops.get_default_graph()
    return apply_op('Assert', condition, data, summarize=summarize, name=name)


def case(pred_fn_pairs, default, exclusive=False, name=None):
    """tf.case via nested cond (reference control_flow_ops.case)."""
    if isinstance(pred_fn_pairs, dict):
        pairs = list(pred_fn_pairs.items())
    else:
        pairs = list(pred_fn_pairs)

    def build(i):
        if i >= len(pairs):
            return default()
        pred, fn = pairs[i]
        return cond(pred, fn, lambda: build(i + 1))

    return build(0)
