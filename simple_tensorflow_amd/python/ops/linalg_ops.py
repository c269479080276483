"""Dense linear algebra ops (analog of reference python/ops/linalg_ops.py;
kernels in csrc/kernels/cpu_linalg.cc)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import apply_op, convert_to_tensor


def _batch_square_shape(t):
    return t._shape


def cholesky(input, name=None):  # noqa: A002
    t = apply_op('Cholesky', convert_to_tensor(input), name=name)
    t.set_shape(_batch_square_shape(t.op.inputs[0]))
    return t


def cholesky_grad(l, grad, name=None):
    return apply_op('CholeskyGrad', l, grad, name=name)


def matrix_determinant(input, name=None):  # noqa: A002
    t = apply_op('MatrixDeterminant', convert_to_tensor(input), name=name)
    inp = t.op.inputs[0]._shape
    if inp is not None:
        t.set_shape(inp[:-2])
    return t


def matrix_inverse(input, adjoint=False, name=None):  # noqa: A002
    t = apply_op('MatrixInverse', convert_to_tensor(input), adjoint=adjoint,
                 name=name)
    t.set_shape(_batch_square_shape(t.op.inputs[0]))
    return t


def matrix_solve(matrix, rhs, adjoint=False, name=None):
    t = apply_op('MatrixSolve', convert_to_tensor(matrix),
                 convert_to_tensor(rhs), adjoint=adjoint, name=name)
    t.set_shape(t.op.inputs[1]._shape)
    return t


def matrix_triangular_solve(matrix, rhs, lower=True, adjoint=False,
                            name=None):
    t = apply_op('MatrixTriangularSolve', convert_to_tensor(matrix),
                 convert_to_tensor(rhs), lower=lower, adjoint=adjoint,
                 name=name)
    t.set_shape(t.op.inputs[1]._shape)
    return t


def matrix_solve_ls(matrix, rhs, l2_regularizer=0.0, fast=True, name=None):
    reg = convert_to_tensor(float(l2_regularizer), dtype=dtypes.float64)
    return apply_op('MatrixSolveLs', convert_to_tensor(matrix),
                    convert_to_tensor(rhs), reg, fast=fast, name=name)


def qr(input, full_matrices=False, name=None):  # noqa: A002
    return apply_op('Qr', convert_to_tensor(input),
                    full_matrices=full_matrices, name=name)


def svd(tensor, full_matrices=False, compute_uv=True, name=None):
    s, u, v = apply_op('Svd', convert_to_tensor(tensor),
                       compute_uv=compute_uv, full_matrices=full_matrices,
                       name=name)
    if compute_uv:
        return s, u, v
    return s


def self_adjoint_eig(tensor, name=None):
    e, v = apply_op('SelfAdjointEigV2', convert_to_tensor(tensor),
                    compute_v=True, name=name)
    return e, v


def self_adjoint_eigvals(tensor, name=None):
    e, _ = apply_op('SelfAdjointEigV2', convert_to_tensor(tensor),
                    compute_v=False, name=name)
    return e


def eye(num_rows, num_columns=None, batch_shape=None, dtype=dtypes.float32,
        name=None):
    import numpy as np
    num_columns = num_columns if num_columns is not None else num_rows
    m = np.eye(num_rows, num_columns, dtype=dtype.as_numpy_dtype
               if hasattr(dtype, 'as_numpy_dtype') else 'float32')
    if batch_shape:
        m = np.broadcast_to(m, tuple(batch_shape) + m.shape).copy()
    return ops.constant(m, name=name)


# ---------------------------------------------------------------------------
# gradients (reference python/ops/linalg_grad.py)
# ---------------------------------------------------------------------------
from simple_tensorflow_amd.python.framework.ops import RegisterGradient, NoGradient  # noqa: E402
from simple_tensorflow_amd.python.ops import array_ops, math_ops  # noqa: E402


@RegisterGradient('MatrixInverse')
def _matrix_inverse_grad(op, grad):
    ainv = op.outputs[0]
    # d(A^-1) = -A^-1 dA A^-1  =>  dA_bar = -A^-T grad A^-T
    return -math_ops.matmul(math_ops.matmul(ainv, grad, transpose_a=True),
                            ainv, transpose_b=True)


@RegisterGradient('MatrixDeterminant')
def _matrix_determinant_grad(op, grad):
    a = op.inputs[0]
    det = op.outputs[0]
    ainv_t = matrix_inverse(a, adjoint=True)
    multipliers = array_ops.reshape(
        grad * det, array_ops.concat([array_ops.shape(det), [1, 1]], 0))
    return multipliers * ainv_t


@RegisterGradient('MatrixSolve')
def _matrix_solve_grad(op, grad):
    a = op.inputs[0]
    adjoint = op.get_attr('adjoint')
    c = op.outputs[0]
    grad_b = matrix_solve(a, grad, adjoint=not adjoint)
    if adjoint:
        grad_a = -math_ops.matmul(c, grad_b, transpose_b=True)
    else:
        grad_a = -math_ops.matmul(grad_b, c, transpose_b=True)
    return grad_a, grad_b


@RegisterGradient('Cholesky')
def _cholesky_grad(op, grad):
    return cholesky_grad(op.outputs[0], grad)


@RegisterGradient('MatrixTriangularSolve')
def _matrix_triangular_solve_grad(op, grad):
    a = op.inputs[0]
    adjoint = op.get_attr('adjoint')
    lower = op.get_attr('lower')
    c = op.outputs[0]
    grad_b = matrix_triangular_solve(a, grad, lower=lower,
                                     adjoint=not adjoint)
    if adjoint:
        grad_a = -math_ops.matmul(c, grad_b, transpose_b=True)
    else:
        grad_a = -math_ops.matmul(grad_b, c, transpose_b=True)
    # project onto the triangle that was actually read
    from simple_tensorflow_amd.python.ops import array_ops as ao
    if lower:
        grad_a = ao.matrix_band_part(grad_a, -1, 0)
    else:
        grad_a = ao.matrix_band_part(grad_a, 0, -1)
    return grad_a, grad_b


for _op in ('CholeskyGrad', 'Qr', 'Svd', 'SelfAdjointEigV2',
            'MatrixSolveLs'):
    NoGradient(_op)
