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
