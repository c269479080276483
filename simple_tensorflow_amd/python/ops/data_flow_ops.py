"""Queues (python wrappers over the FIFOQueue/RandomShuffleQueue kernels —
analog of reference python/ops/data_flow_ops.py)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import apply_op, convert_to_tensor


class QueueBase(object):
    def __init__(self, handle, dtypes_list, shapes, names=None):
        self._handle = handle
        self._dtypes = [dtypes.as_dtype(d) for d in dtypes_list]
        self._shapes = shapes

    @property
    def queue_ref(self):
        return self._handle

    @property
    def dtypes(self):
        return self._dtypes

    def enqueue(self, vals, name=None):
        if not isinstance(vals, (list, tuple)):
            vals = [vals]
        vals = [convert_to_tensor(v, dtype=d)
                for v, d in zip(vals, self._dtypes)]
        return apply_op('QueueEnqueue', self._handle, vals, name=name)

    def enqueue_many(self, vals, name=None):
        if not isinstance(vals, (list, tuple)):
            vals = [vals]
        vals = [convert_to_tensor(v, dtype=d)
                for v, d in zip(vals, self._dtypes)]
        return apply_op('QueueEnqueueMany', self._handle, vals, name=name)

    def _set_shapes(self, outs, extra_batch=None):
        outs = list(outs) if isinstance(outs, tuple) else [outs]
        if self._shapes:
            for t, sh in zip(outs, self._shapes):
                dims = list(sh) if sh is not None else None
                if dims is not None and extra_batch is not None:
                    dims = [extra_batch] + dims
                t.set_shape(dims)
        return outs

    def dequeue(self, name=None):
        outs = apply_op('QueueDequeue', self._handle,
                        component_types=self._dtypes, name=name)
        outs = self._set_shapes(outs)
        return outs[0] if len(outs) == 1 else outs

    def dequeue_many(self, n, name=None):
        outs = apply_op('QueueDequeueMany', self._handle,
                        convert_to_tensor(n, dtype=dtypes.int32),
                        component_types=self._dtypes, name=name)
        outs = self._set_shapes(outs, extra_batch=n if isinstance(n, int)
                                else None)
        return outs[0] if len(outs) == 1 else outs

    def close(self, cancel_pending_enqueues=False, name=None):
        return apply_op('QueueClose', self._handle,
                        cancel_pending_enqueues=cancel_pending_enqueues,
                        name=name)

    def size(self, name=None):
        return apply_op('QueueSize', self._handle, name=name)


class FIFOQueue(QueueBase):
    def __init__(self, capacity, dtypes_list=None, shapes=None, names=None,
                 shared_name=None, name='fifo_queue', dtypes=None):
        dtypes_list = dtypes_list if dtypes_list is not None else dtypes
        if not isinstance(dtypes_list, (list, tuple)):
            dtypes_list = [dtypes_list]
        handle = apply_op('FIFOQueue', component_types=dtypes_list,
                          capacity=capacity, name=name)
        super().__init__(handle, dtypes_list, shapes)


class RandomShuffleQueue(QueueBase):
    def __init__(self, capacity, min_after_dequeue, dtypes_list=None,
                 shapes=None, names=None, seed=None, shared_name=None,
                 name='random_shuffle_queue', dtypes=None):
        dtypes_list = dtypes_list if dtypes_list is not None else dtypes
        if not isinstance(dtypes_list, (list, tuple)):
            dtypes_list = [dtypes_list]
        handle = apply_op('RandomShuffleQueue', component_types=dtypes_list,
                          capacity=capacity,
                          min_after_dequeue=min_after_dequeue,
                          seed=seed or 0, name=name)
        super().__init__(handle, dtypes_list, shapes)


class PaddingFIFOQueue(FIFOQueue):
    pass


class ConditionalAccumulator(object):
    """Aggregates gradients conditionally on freshness (reference
    python/ops/data_flow_ops.py ConditionalAccumulator; kernels in
    csrc/kernels/cpu_accumulator.cc)."""

    def __init__(self, dtype, shape=None, shared_name=None,
                 name='conditional_accumulator'):
        from simple_tensorflow_amd.python.framework import dtypes as _dt
        self._dtype = _dt.as_dtype(dtype)
        self._shape = list(shape) if shape is not None else None
        self._handle = apply_op(
            'ConditionalAccumulator', dtype=int(self._dtype),
            shape=self._shape or [], shared_name=shared_name or '',
            name=name)

    @property
    def accumulator_ref(self):
        return self._handle

    def apply_grad(self, grad, local_step=0, name=None):
        local_step = ops.convert_to_tensor(local_step, dtype=dtypes.int64)
        grad = ops.convert_to_tensor(grad, dtype=self._dtype)
        return apply_op('AccumulatorApplyGradient', self._handle, local_step,
                        grad, name=name)

    def take_grad(self, num_required, name=None):
        num_required = ops.convert_to_tensor(num_required,
                                             dtype=dtypes.int32)
        out = apply_op('AccumulatorTakeGradient', self._handle, num_required,
                       dtype=int(self._dtype), name=name)
        if self._shape is not None:
            out.set_shape(self._shape)
        return out

    def set_global_step(self, new_global_step, name=None):
        new_global_step = ops.convert_to_tensor(new_global_step,
                                                dtype=dtypes.int64)
        return apply_op('AccumulatorSetGlobalStep', self._handle,
                        new_global_step, name=name)

    def num_accumulated(self, name=None):
        return apply_op('AccumulatorNumAccumulated', self._handle, name=name)
