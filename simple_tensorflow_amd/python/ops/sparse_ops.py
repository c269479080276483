"""SparseTensor + sparse_to_dense (reference python/framework/sparse_tensor.py
+ python/ops/sparse_ops.py subset)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import apply_op, convert_to_tensor


class SparseTensor(object):
    def __init__(self, indices, values, dense_shape):
        self.indices = convert_to_tensor(indices, dtype=dtypes.int64)
        self.values = convert_to_tensor(values)
        self.dense_shape = convert_to_tensor(dense_shape,
                                             dtype=dtypes.int64)

    @property
    def dtype(self):
        return self.values.dtype

    @property
    def shape(self):
        return self.dense_shape

    def get_shape(self):
        return self.dense_shape


SparseTensorValue = tuple


def sparse_to_dense(sparse_indices, output_shape, sparse_values,
                    default_value=0, validate_indices=True, name=None):
    sparse_indices = convert_to_tensor(sparse_indices, dtype=dtypes.int64)
    output_shape = convert_to_tensor(output_shape, dtype=dtypes.int64)
    sparse_values = convert_to_tensor(sparse_values)
    default_value = convert_to_tensor(default_value,
                                      dtype=sparse_values.dtype)
    return apply_op('SparseToDense', sparse_indices, output_shape,
                    sparse_values, default_value,
                    validate_indices=validate_indices, name=name)


def sparse_tensor_to_dense(sp, default_value=0, validate_indices=True,
                           name=None):
    return sparse_to_dense(sp.indices, sp.dense_shape, sp.values,
                           default_value, validate_indices, name)


def sparse_tensor_dense_matmul(sp_a, b, adjoint_a=False, adjoint_b=False,
                               name=None):
    """out[i, :] = sum_k A[i, k] * B[k, :] for COO A (reference
    sparse_ops.sparse_tensor_dense_matmul). Formulated as gather rows of B
    by A's column indices, scale by A's values, and segment-sum by A's row
    indices — all three run on the GPU gather/scatter kernels."""
    if adjoint_a or adjoint_b:
        raise NotImplementedError(
            'sparse_tensor_dense_matmul adjoint_a/adjoint_b')
    from simple_tensorflow_amd.python.ops import array_ops, math_ops
    g = ops.get_default_graph()
    with g.name_scope(name or 'SparseTensorDenseMatMul'):
        rows = math_ops.cast(sp_a.indices[:, 0], dtypes.int32)
        cols = math_ops.cast(sp_a.indices[:, 1], dtypes.int32)
        gathered = array_ops.gather(convert_to_tensor(b), cols)
        scaled = gathered * array_ops.reshape(sp_a.values, [-1, 1])
        n_rows = math_ops.cast(sp_a.dense_shape[0], dtypes.int32)
        return array_ops.unsorted_segment_sum(scaled, rows, n_rows)


def sparse_add(a, b, thresh=0.0):
    """SparseTensor + SparseTensor -> SparseTensor (reference sparse_ops.py
    sparse_add; kernel csrc/kernels/cpu_sparse.cc)."""
    t = convert_to_tensor(float(thresh), dtype=dtypes.float32)
    idx, vals, shape = apply_op('SparseAdd', a.indices, a.values,
                                a.dense_shape, b.indices, b.values,
                                b.dense_shape, t)
    return SparseTensor(idx, vals, shape)


def sparse_tensor_dense_add(sp, dense):
    return apply_op('SparseTensorDenseAdd', sp.indices, sp.values,
                    sp.dense_shape, convert_to_tensor(dense))


def sparse_reorder(sp, name=None):
    idx, vals = apply_op('SparseReorder', sp.indices, sp.values,
                         sp.dense_shape, name=name)
    return SparseTensor(idx, vals, sp.dense_shape)


def sparse_reduce_sum(sp, axis=None, keep_dims=False,
                      reduction_axes=None):
    import numpy as np
    if axis is None:
        axis = reduction_axes
    if axis is None:
        nd = None
        # reduce over all dims: pass every axis index
        axis = list(range(int(sp.indices._shape[1])))
    axes = np.asarray(axis, np.int32).reshape(-1)
    return apply_op('SparseReduceSum', sp.indices, sp.values, sp.dense_shape,
                    convert_to_tensor(axes), keep_dims=keep_dims)


def sparse_concat(axis, sp_inputs, name=None):
    idx, vals, shape = apply_op(
        'SparseConcat', [sp.indices for sp in sp_inputs],
        [sp.values for sp in sp_inputs],
        [sp.dense_shape for sp in sp_inputs], concat_dim=axis, name=name)
    return SparseTensor(idx, vals, shape)


def sparse_retain(sp, to_retain):
    """Keep only entries where to_retain is True (python-level filter as in
    the reference's sparse_ops.sparse_retain)."""
    from simple_tensorflow_amd.python.ops import array_ops
    keep = array_ops.where(convert_to_tensor(to_retain))
    keep = array_ops.reshape(keep, [-1])
    from simple_tensorflow_amd.python.framework import dtypes as _dt
    from simple_tensorflow_amd.python.ops import math_ops
    keep32 = math_ops.cast(keep, _dt.int32)
    new_idx = array_ops.gather(sp.indices, keep32)
    new_vals = array_ops.gather(sp.values, keep32)
    return SparseTensor(new_idx, new_vals, sp.dense_shape)
