"""tf.einsum (reference python/ops/special_math_ops.py): parsed into
transpose / reshape / batch_matmul / reduce_sum compositions. Supports 1-
and 2-operand expressions without ellipsis or repeated labels per operand;
shapes must be statically known (they always are in this framework's
models)."""
import numpy as np

from simple_tensorflow_amd.python.framework.ops import convert_to_tensor
from simple_tensorflow_amd.python.ops import array_ops, math_ops


def _parse(equation, n_inputs):
    equation = equation.replace(' ', '')
    if '...' in equation:
        raise ValueError('einsum: ellipsis not supported')
    if '->' in equation:
        lhs, out = equation.split('->')
    else:
        lhs = equation
        # implicit output: labels appearing exactly once, sorted
        counts = {}
        for ch in lhs.replace(',', ''):
            counts[ch] = counts.get(ch, 0) + 1
        out = ''.join(sorted(c for c, n in counts.items() if n == 1))
    terms = lhs.split(',')
    if len(terms) != n_inputs:
        raise ValueError('einsum: %d inputs for equation %r' %
                         (n_inputs, equation))
    for t in terms:
        if len(set(t)) != len(t):
            raise ValueError('einsum: repeated label within one operand '
                             'not supported (%r)' % t)
    return terms, out


def _static_shape(t):
    if t._shape is None or any(d is None for d in t._shape):
        raise ValueError('einsum requires statically known shapes')
    return [int(d) for d in t._shape]


def einsum(equation, *inputs, **kwargs):
    inputs = [convert_to_tensor(x) for x in inputs]
    terms, out = _parse(equation, len(inputs))

    if len(inputs) == 1:
        a, term = inputs[0], terms[0]
        shape = _static_shape(a)
        # permute output labels first, reduced labels last
        reduced = [c for c in term if c not in out]
        perm = [term.index(c) for c in out] + [term.index(c) for c in reduced]
        res = array_ops.transpose(a, perm) if perm != list(range(len(perm))) \
            else a
        if reduced:
            axes = list(range(len(out), len(term)))
            res = math_ops.reduce_sum(res, axes)
        return res

    a, b = inputs
    ta, tb = terms
    sa, sb = _static_shape(a), _static_shape(b)
    dim = {}
    for t, s in ((ta, sa), (tb, sb)):
        for c, d in zip(t, s):
            if c in dim and dim[c] != d:
                raise ValueError('einsum: dimension mismatch for %r' % c)
            dim[c] = d
    batch = [c for c in ta if c in tb and c in out]
    contract = [c for c in ta if c in tb and c not in out]
    free_a = [c for c in ta if c not in tb]
    free_b = [c for c in tb if c not in ta]
    for c in free_a + free_b:
        if c not in out:
            raise ValueError('einsum: label %r reduced from a single '
                             'operand; write it as a 1-operand reduction '
                             'first' % c)

    def prep(x, term, lead, tail):
        perm = [term.index(c) for c in batch + lead + tail]
        if perm != list(range(len(perm))):
            x = array_ops.transpose(x, perm)
        bshape = [dim[c] for c in batch]
        m = int(np.prod([dim[c] for c in lead])) if lead else 1
        n = int(np.prod([dim[c] for c in tail])) if tail else 1
        return array_ops.reshape(x, bshape + [m, n])

    a3 = prep(a, ta, free_a, contract)      # [B..., F1, C]
    b3 = prep(b, tb, contract, free_b)      # [B..., C, F2]
    if batch:
        prod = math_ops.batch_matmul(a3, b3)
    else:
        prod = math_ops.matmul(array_ops.reshape(a3, _static_shape(a3)[-2:]),
                               array_ops.reshape(b3, _static_shape(b3)[-2:]))
    interm = batch + free_a + free_b
    res = array_ops.reshape(prod, [dim[c] for c in interm])
    perm = [interm.index(c) for c in out]
    if perm != list(range(len(perm))):
        res = array_ops.transpose(res, perm)
    return res
