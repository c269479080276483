"""tf.TensorArray (reference python/ops/tensor_array_ops.py TensorArray:54;
kernels in csrc/kernels/tensor_array.cc use session-scoped string handles)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import apply_op, convert_to_tensor
from simple_tensorflow_amd.python.ops import array_ops, math_ops


class TensorArray(object):
    def __init__(self, dtype, size=None, dynamic_size=False,
                 clear_after_read=True, tensor_array_name=None, handle=None,
                 flow=None, infer_shape=True, element_shape=None, name=None):
        self._dtype = dtypes.as_dtype(dtype)
        self._infer_shape = infer_shape
        self._element_shape = element_shape
        if handle is not None:
            self._handle = handle
            self._flow = flow
        else:
            if size is None:
                raise ValueError('size must be provided')
            size_t = convert_to_tensor(size, dtype=dtypes.int32)
            self._handle, self._flow = apply_op(
                'TensorArrayV3', size_t, dtype=int(self._dtype),
                dynamic_size=dynamic_size, clear_after_read=clear_after_read,
                tensor_array_name=tensor_array_name or '',
                name=name or 'TensorArray')

    @property
    def dtype(self):
        return self._dtype

    @property
    def handle(self):
        return self._handle

    @property
    def flow(self):
        return self._flow

    def _with_flow(self, flow):
        ta = TensorArray(self._dtype, handle=self._handle, flow=flow,
                         infer_shape=self._infer_shape,
                         element_shape=self._element_shape)
        return ta

    def identity(self):
        return self._with_flow(self._flow)

    def grad(self, source, flow=None):
        flow = flow if flow is not None else self._flow
        g_handle, g_flow = apply_op('TensorArrayGradV3', self._handle, flow,
                                    source=source)
        return TensorArray(self._dtype, handle=g_handle, flow=g_flow)

    def read(self, index, name=None):
        index = convert_to_tensor(index, dtype=dtypes.int32)
        value = apply_op('TensorArrayReadV3', self._handle, index,
                         self._flow, dtype=int(self._dtype), name=name)
        if self._element_shape is not None:
            value.set_shape(self._element_shape)
        return value

    def write(self, index, value, name=None):
        index = convert_to_tensor(index, dtype=dtypes.int32)
        value = convert_to_tensor(value, dtype=self._dtype)
        flow = apply_op('TensorArrayWriteV3', self._handle, index, value,
                        self._flow, name=name)
        if self._infer_shape and self._element_shape is None and \
                value._shape is not None:
            self._element_shape = list(value._shape)
        return self._with_flow(flow)

    def size(self, name=None):
        return apply_op('TensorArraySizeV3', self._handle, self._flow,
                        name=name)

    def stack(self, name=None):
        return self.gather(math_ops.range(0, self.size()), name=name)

    def gather(self, indices, name=None):
        indices = convert_to_tensor(indices, dtype=dtypes.int32)
        value = apply_op('TensorArrayGatherV3', self._handle, indices,
                         self._flow, dtype=int(self._dtype), name=name)
        if self._element_shape is not None:
            value.set_shape([None] + list(self._element_shape))
        return value

    def unstack(self, value, name=None):
        value = convert_to_tensor(value, dtype=self._dtype)
        num = value._shape[0] if value._shape is not None else None
        if num is None:
            indices = math_ops.range(0, array_ops.shape(value)[0])
        else:
            indices = math_ops.range(0, num)
        return self.scatter(indices, value, name=name)

    def scatter(self, indices, value, name=None):
        indices = convert_to_tensor(indices, dtype=dtypes.int32)
        value = convert_to_tensor(value, dtype=self._dtype)
        flow = apply_op('TensorArrayScatterV3', self._handle, indices, value,
                        self._flow, name=name)
        return self._with_flow(flow)

    def close(self, name=None):
        return apply_op('TensorArrayCloseV3', self._handle, name=name)
