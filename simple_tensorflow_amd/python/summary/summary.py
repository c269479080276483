"""tf.summary ops (analog of reference python/summary/summary.py)."""
from simple_tensorflow_amd.python.framework import dtypes, ops
from simple_tensorflow_amd.python.framework.ops import apply_op, convert_to_tensor


def scalar(name, tensor, collections=None):
    g = ops.get_default_graph()
    t = apply_op('ScalarSummary', ops.constant(name),
                 convert_to_tensor(tensor), name=name.replace(' ', '_'))
    for c in (collections or [ops.GraphKeys.SUMMARIES]):
        g.add_to_collection(c, t)
    return t


def histogram(name, values, collections=None):
    g = ops.get_default_graph()
    t = apply_op('HistogramSummary', ops.constant(name),
                 convert_to_tensor(values), name=name.replace(' ', '_'))
    for c in (collections or [ops.GraphKeys.SUMMARIES]):
        g.add_to_collection(c, t)
    return t


def merge(inputs, collections=None, name=None):
    return apply_op('MergeSummary', list(inputs), name=name)


def merge_all(key=ops.GraphKeys.SUMMARIES):
    summaries = ops.get_default_graph().get_collection(key)
    if not summaries:
        return None
    return merge(summaries)
