"""Spectral ops: FFT family + complex accessors (reference
python/ops/spectral_ops.py + math_ops complex section; kernels in
csrc/kernels/cpu_fft.cc)."""
import numpy as np

from simple_tensorflow_amd.python.framework import dtypes
from simple_tensorflow_amd.python.framework.ops import (
    NoGradient, RegisterGradient, apply_op, convert_to_tensor)


def _c(x):
    return convert_to_tensor(x, dtype=dtypes.complex64)


def _same_shape(t):
    t.set_shape(t.op.inputs[0]._shape)
    return t


def fft(input, name=None):  # noqa: A002
    return _same_shape(apply_op('FFT', _c(input), name=name))


def ifft(input, name=None):  # noqa: A002
    return _same_shape(apply_op('IFFT', _c(input), name=name))


def fft2d(input, name=None):  # noqa: A002
    return _same_shape(apply_op('FFT2D', _c(input), name=name))


def ifft2d(input, name=None):  # noqa: A002
    return _same_shape(apply_op('IFFT2D', _c(input), name=name))


def fft3d(input, name=None):  # noqa: A002
    return _same_shape(apply_op('FFT3D', _c(input), name=name))


def ifft3d(input, name=None):  # noqa: A002
    return _same_shape(apply_op('IFFT3D', _c(input), name=name))


def _length(fft_length):
    return convert_to_tensor(np.asarray(fft_length, np.int32))


def rfft(input_tensor, fft_length=None, name=None):
    t = convert_to_tensor(input_tensor, dtype=dtypes.float32)
    if fft_length is None:
        fft_length = [t._shape[-1]]
    return apply_op('RFFT', t, _length(fft_length), name=name)


def irfft(input_tensor, fft_length=None, name=None):
    t = _c(input_tensor)
    if fft_length is None:
        fft_length = [2 * (t._shape[-1] - 1)]
    return apply_op('IRFFT', t, _length(fft_length), name=name)


def rfft2d(input_tensor, fft_length=None, name=None):
    t = convert_to_tensor(input_tensor, dtype=dtypes.float32)
    if fft_length is None:
        fft_length = [t._shape[-2], t._shape[-1]]
    return apply_op('RFFT2D', t, _length(fft_length), name=name)


def irfft2d(input_tensor, fft_length=None, name=None):
    t = _c(input_tensor)
    if fft_length is None:
        fft_length = [t._shape[-2], 2 * (t._shape[-1] - 1)]
    return apply_op('IRFFT2D', t, _length(fft_length), name=name)


def complex(real, imag, name=None):  # noqa: A001
    return _same_shape(apply_op(
        'Complex', convert_to_tensor(real, dtype=dtypes.float32),
        convert_to_tensor(imag, dtype=dtypes.float32), name=name))


def real(input, name=None):  # noqa: A002
    return _same_shape(apply_op('Real', _c(input), name=name))


def imag(input, name=None):  # noqa: A002
    return _same_shape(apply_op('Imag', _c(input), name=name))


def conj(input, name=None):  # noqa: A002
    return _same_shape(apply_op('Conj', _c(input), name=name))


def complex_abs(input, name=None):  # noqa: A002
    return _same_shape(apply_op('ComplexAbs', _c(input), name=name))


# ---------------------------------------------------------------------------
# gradients (reference python/ops/spectral_grad + math_grad complex entries)
# ---------------------------------------------------------------------------
def _scale(t, s):
    # scalar * complex without requiring complex binary kernels
    from simple_tensorflow_amd.python.ops import math_ops
    return complex(math_ops.multiply(real(t), s), math_ops.multiply(imag(t), s))


def _fft_size(op, rank):
    n = 1
    shape = op.inputs[0]._shape
    for d in shape[-rank:]:
        n *= int(d)
    return float(n)


@RegisterGradient('FFT')
def _fft_grad(op, grad):
    return _scale(ifft(grad), _fft_size(op, 1))


@RegisterGradient('IFFT')
def _ifft_grad(op, grad):
    return _scale(fft(grad), 1.0 / _fft_size(op, 1))


@RegisterGradient('FFT2D')
def _fft2d_grad(op, grad):
    return _scale(ifft2d(grad), _fft_size(op, 2))


@RegisterGradient('IFFT2D')
def _ifft2d_grad(op, grad):
    return _scale(fft2d(grad), 1.0 / _fft_size(op, 2))


@RegisterGradient('Conj')
def _conj_grad(op, grad):
    return conj(grad)


@RegisterGradient('Real')
def _real_grad(op, grad):
    from simple_tensorflow_amd.python.ops import array_ops
    return complex(grad, array_ops.zeros_like(grad))


@RegisterGradient('Imag')
def _imag_grad(op, grad):
    from simple_tensorflow_amd.python.ops import array_ops
    return complex(array_ops.zeros_like(grad), grad)


@RegisterGradient('Complex')
def _complex_grad(op, grad):
    return real(grad), imag(grad)


for _op in ('FFT3D', 'IFFT3D', 'RFFT', 'IRFFT', 'RFFT2D', 'IRFFT2D',
            'ComplexAbs'):
    NoGradient(_op)
