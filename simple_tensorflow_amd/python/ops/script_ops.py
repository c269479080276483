"""tf.py_func — run a python callable inside the graph (reference
python/ops/script_ops.py py_func + python/lib/core/py_func.cc; the bridge
kernel lives in csrc/pybind/module.cc PyFuncOp)."""
import uuid

from simple_tensorflow_amd import _core
from simple_tensorflow_amd.python.framework import dtypes
from simple_tensorflow_amd.python.framework.ops import apply_op, convert_to_tensor


def py_func(func, inp, Tout, stateful=True, name=None):
    single = not isinstance(Tout, (list, tuple))
    touts = [Tout] if single else list(Tout)
    touts = [int(dtypes.as_dtype(t)) for t in touts]
    token = 'pyfunc_%s' % uuid.uuid4().hex
    _core.register_py_func(token, func)
    tensors = [convert_to_tensor(x) for x in inp]
    out = apply_op('PyFunc' if stateful else 'PyFuncStateless', tensors,
                   token=token, Tout=touts, name=name)
    if single:
        return out if not isinstance(out, tuple) else out[0]
    return list(out) if isinstance(out, tuple) else [out]
