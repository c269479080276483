"""replica_device_setter (analog of reference python/training/
device_setter.py:124): round-robin placement of variables on ps tasks.
With the MI355X single-node RCCL path this is API parity; the returned device
chooser emits /job:ps round-robin for variable ops and the worker device for
everything else."""

_VARIABLE_OPS = {'Variable', 'VariableV2'}


class _RoundRobinStrategy(object):
    def __init__(self, num_tasks):
        self._num_tasks = num_tasks
        self._next = 0

    def __call__(self, op):
        t = self._next
        self._next = (self._next + 1) % self._num_tasks
        return t


def replica_device_setter(ps_tasks=0, ps_device='/job:ps',
                          worker_device='/job:worker', merge_devices=True,
                          cluster=None, ps_ops=None, ps_strategy=None):
    if cluster is not None:
        ps_tasks = cluster.num_tasks('ps') if 'ps' in cluster.jobs() else 0
    if ps_tasks == 0:
        return lambda op: worker_device
    ps_ops = ps_ops or _VARIABLE_OPS
    strategy = ps_strategy or _RoundRobinStrategy(ps_tasks)

    def chooser(op):
        node_op = op.type if hasattr(op, 'type') else ''
        if node_op in ps_ops:
            return '%s/task:%d' % (ps_device, strategy(op))
        return worker_device

    return chooser
