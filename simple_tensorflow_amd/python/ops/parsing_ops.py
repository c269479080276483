"""parse_example / parse_single_example (reference python/ops/parsing_ops.py
+ core/kernels/example_parsing_ops.cc): host-side wire parsing through the
py_func bridge — exactly where the reference's CPU parse kernels run."""
import numpy as np

from simple_tensorflow_amd.python.framework import dtypes
from simple_tensorflow_amd.python.framework.ops import convert_to_tensor
from simple_tensorflow_amd.python.lib.example_pb import parse_example_bytes
from simple_tensorflow_amd.python.ops import script_ops


class FixedLenFeature(object):
    def __init__(self, shape, dtype, default_value=None):
        self.shape = list(shape)
        self.dtype = dtypes.as_dtype(dtype)
        self.default_value = default_value


class VarLenFeature(object):
    def __init__(self, dtype):
        self.dtype = dtypes.as_dtype(dtype)


def _extract(parsed, key, spec):
    val = parsed.get(key)
    if val is None:
        if isinstance(spec, FixedLenFeature) and \
                spec.default_value is not None:
            return np.asarray(spec.default_value,
                              spec.dtype.as_numpy_dtype).reshape(spec.shape)
        raise ValueError('feature %r missing and no default' % key)
    if spec.dtype == dtypes.string:
        return val  # list of bytes
    arr = np.asarray(val, spec.dtype.as_numpy_dtype)
    if isinstance(spec, FixedLenFeature):
        return arr.reshape(spec.shape)
    return arr


def parse_single_example(serialized, features, name=None,
                         example_names=None):
    keys = sorted(features)
    touts = []
    for k in keys:
        sp = features[k]
        if isinstance(sp, VarLenFeature):
            raise ValueError('parse_single_example: VarLenFeature not '
                             'supported (use FixedLenFeature)')
        touts.append(sp.dtype)

    def _parse(blob):
        b = blob if isinstance(blob, bytes) else bytes(blob)
        parsed = parse_example_bytes(b)
        outs = []
        for k in keys:
            v = _extract(parsed, k, features[k])
            if features[k].dtype == dtypes.string:
                v = v[0] if v else b''
            outs.append(v)
        return tuple(outs) if len(outs) > 1 else outs[0]

    results = script_ops.py_func(_parse, [convert_to_tensor(serialized)],
                                 touts if len(touts) > 1 else touts[0],
                                 name=name)
    if not isinstance(results, (list, tuple)):
        results = [results]
    out = {}
    for k, t, in zip(keys, results):
        sp = features[k]
        if isinstance(sp, FixedLenFeature) and sp.dtype != dtypes.string:
            t.set_shape(sp.shape)
        out[k] = t
    return out


def parse_example(serialized, features, name=None, example_names=None):
    """Batch variant: serialized is a 1-D string tensor; FixedLenFeature
    outputs gain a leading batch dimension."""
    keys = sorted(features)
    touts = [features[k].dtype for k in keys]

    def _parse(blobs):
        batches = [[] for _ in keys]
        for blob in blobs:
            b = blob if isinstance(blob, bytes) else bytes(blob)
            parsed = parse_example_bytes(b)
            for i, k in enumerate(keys):
                batches[i].append(_extract(parsed, k, features[k]))
        outs = [np.stack(b) for b in batches]
        return tuple(outs) if len(outs) > 1 else outs[0]

    results = script_ops.py_func(_parse, [convert_to_tensor(serialized)],
                                 touts if len(touts) > 1 else touts[0],
                                 name=name)
    if not isinstance(results, (list, tuple)):
        results = [results]
    out = {}
    for k, t in zip(keys, results):
        sp = features[k]
        if isinstance(sp, FixedLenFeature):
            t.set_shape([None] + sp.shape)
        out[k] = t
    return out
