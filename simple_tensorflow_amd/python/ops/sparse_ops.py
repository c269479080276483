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
