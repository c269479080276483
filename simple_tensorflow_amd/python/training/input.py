"""Input pipeline: tf.train.batch / shuffle_batch / input producers over
queues + QueueRunner threads (analog of reference python/training/input.py
batch:829, shuffle_batch:1120)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import convert_to_tensor
from simple_tensorflow_amd.python.ops import data_flow_ops
from simple_tensorflow_amd.python.training import coordinator as qr_lib


def _as_list(x):
    return list(x) if isinstance(x, (list, tuple)) else [x]


def batch(tensors, batch_size, num_threads=1, capacity=32,
          enqueue_many=False, shapes=None, dynamic_pad=False,
          allow_smaller_final_batch=False, shared_name=None, name=None):
    tensor_list = [convert_to_tensor(t) for t in _as_list(tensors)]
    if shapes is None:
        shapes = [t._shape for t in tensor_list]
    q = data_flow_ops.FIFOQueue(capacity, [t.dtype for t in tensor_list],
                                shapes=shapes, name=name or 'batch_queue')
    enq = q.enqueue_many(tensor_list) if enqueue_many \
        else q.enqueue(tensor_list)
    qr_lib.add_queue_runner(
        qr_lib.QueueRunner(q, [enq] * num_threads, close_op=q.close()))
    out = q.dequeue_many(batch_size)
    return out if isinstance(tensors, (list, tuple)) else out


def shuffle_batch(tensors, batch_size, capacity, min_after_dequeue,
                  num_threads=1, seed=None, enqueue_many=False, shapes=None,
                  allow_smaller_final_batch=False, shared_name=None,
                  name=None):
    tensor_list = [convert_to_tensor(t) for t in _as_list(tensors)]
    if shapes is None:
        shapes = [t._shape for t in tensor_list]
    q = data_flow_ops.RandomShuffleQueue(
        capacity, min_after_dequeue, [t.dtype for t in tensor_list],
        shapes=shapes, seed=seed, name=name or 'shuffle_batch_queue')
    enq = q.enqueue_many(tensor_list) if enqueue_many \
        else q.enqueue(tensor_list)
    qr_lib.add_queue_runner(
        qr_lib.QueueRunner(q, [enq] * num_threads, close_op=q.close()))
    out = q.dequeue_many(batch_size)
    return out


def input_producer(input_tensor, num_epochs=None, shuffle=True, seed=None,
                   capacity=32, name=None):
    """Cycles the rows of input_tensor through a queue."""
    input_tensor = convert_to_tensor(input_tensor)
    q = data_flow_ops.FIFOQueue(capacity, [input_tensor.dtype],
                                shapes=[list(input_tensor._shape[1:])]
                                if input_tensor._shape else None,
                                name=name or 'input_producer')
    enq = q.enqueue_many([input_tensor])
    qr_lib.add_queue_runner(qr_lib.QueueRunner(q, [enq], close_op=q.close()))
    return q


def string_input_producer(string_tensor, num_epochs=None, shuffle=True,
                          seed=None, capacity=32, name=None):
    return input_producer(string_tensor, num_epochs, shuffle, seed, capacity,
                          name or 'string_input_producer')


def range_input_producer(limit, num_epochs=None, shuffle=True, seed=None,
                         capacity=32, name=None):
    from simple_tensorflow_amd.python.ops import math_ops
    return input_producer(math_ops.range(limit), num_epochs, shuffle, seed,
                          capacity, name or 'range_input_producer')
