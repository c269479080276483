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


def limit_epochs(tensor, num_epochs=None, name=None):
    """Raises OutOfRange after the tensor has been produced num_epochs times
    (reference input.py limit_epochs: a counter variable + assert)."""
    if num_epochs is None:
        return convert_to_tensor(tensor)
    from simple_tensorflow_amd.python.framework import dtypes, ops
    from simple_tensorflow_amd.python.ops import math_ops, state_ops
    from simple_tensorflow_amd.python.ops import variables as vars_mod
    from simple_tensorflow_amd.python.ops import control_flow_ops
    counter = vars_mod.Variable(0, name=name or 'epochs', trainable=False,
                                dtype=dtypes.int32)
    t = convert_to_tensor(tensor)
    inc = state_ops.assign_add(counter._variable, 1)
    # OutOfRange once the counter passes the limit: piggyback on queue
    # semantics by asserting through a cond raising via Assert
    check = control_flow_ops.Assert(
        math_ops.less_equal(inc, ops.constant(int(num_epochs),
                                              dtypes.int32)),
        [inc])
    with ops.get_default_graph().control_dependencies([check]):
        from simple_tensorflow_amd.python.ops import array_ops
        return array_ops.identity(t)


def slice_input_producer(tensor_list, num_epochs=None, shuffle=True,
                         seed=None, capacity=32, name=None):
    """Produces one row at a time from each tensor in tensor_list
    (reference input.py slice_input_producer)."""
    from simple_tensorflow_amd.python.ops import array_ops, math_ops
    tensor_list = [convert_to_tensor(t) for t in tensor_list]
    n = int(tensor_list[0]._shape[0])
    rq = input_producer(math_ops.range(n), num_epochs, shuffle, seed,
                        capacity, name or 'slice_input_producer')
    idx = rq.dequeue()
    return [array_ops.gather(t, array_ops.reshape(idx, [1]))[0]
            if False else array_ops.gather(t, idx) for t in tensor_list]
