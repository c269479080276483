"""DType system — enum values wire-compatible with the reference's
types.proto (reference: tensorflow/python/framework/dtypes.py)."""
import numpy as np


class DType(object):
    def __init__(self, enum, name, np_dtype, is_floating, is_integer):
        self._enum = enum
        self.name = name
        self.as_numpy_dtype = np_dtype
        self.is_floating = is_floating
        self.is_integer = is_integer

    @property
    def as_datatype_enum(self):
        return self._enum

    @property
    def base_dtype(self):
        return self

    @property
    def is_ref_dtype(self):
        return False

    def __int__(self):
        return self._enum

    def __eq__(self, other):
        if other is None:
            return False
        try:
            return self._enum == as_dtype(other)._enum
        except (TypeError, ValueError):
            return False

    def __ne__(self, other):
        return not self.__eq__(other)

    def __hash__(self):
        return self._enum

    def __repr__(self):
        return 'tf.' + self.name


float32 = DType(1, 'float32', np.float32, True, False)
float64 = DType(2, 'float64', np.float64, True, False)
int32 = DType(3, 'int32', np.int32, False, True)
uint8 = DType(4, 'uint8', np.uint8, False, True)
int16 = DType(5, 'int16', np.int16, False, True)
int8 = DType(6, 'int8', np.int8, False, True)
string = DType(7, 'string', np.object_, False, False)
int64 = DType(9, 'int64', np.int64, False, True)
bool = DType(10, 'bool', np.bool_, False, False)
bfloat16 = DType(14, 'bfloat16', np.uint16, True, False)
uint16 = DType(17, 'uint16', np.uint16, False, True)
float16 = DType(19, 'float16', np.float16, True, False)
complex64 = DType(8, 'complex64', np.complex64, False, False)
complex128 = DType(18, 'complex128', np.complex128, False, False)
half = float16
double = float64

_ALL = [float32, float64, int32, uint8, int16, int8, string, int64, bool,
        bfloat16, uint16, float16, complex64, complex128]
_BY_ENUM = {d._enum: d for d in _ALL}
_BY_NAME = {d.name: d for d in _ALL}
_BY_NAME.update({'float': float32, 'double': float64, 'half': float16})


def as_dtype(v):
    if isinstance(v, DType):
        return v
    if isinstance(v, int):
        return _BY_ENUM[v]
    if isinstance(v, str):
        return _BY_NAME[v]
    if v is float:
        return float32
    if v is int:
        return int32
    npdt = np.dtype(v)
    for d in _ALL:
        if d is bfloat16 or d is string:
            continue
        if np.dtype(d.as_numpy_dtype) == npdt:
            return d
    raise TypeError('Cannot convert %r to a DType' % (v,))
