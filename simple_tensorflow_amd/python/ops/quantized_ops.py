"""Quantized ops (reference array_ops quantize section; kernels in
csrc/kernels/cpu_quantized.cc — quint8/qint32 carried as uint8/int32)."""
from simple_tensorflow_amd.python.framework import dtypes
from simple_tensorflow_amd.python.framework.ops import (
    NoGradient, apply_op, convert_to_tensor)


def quantize_v2(input, min_range, max_range, T=dtypes.uint8,  # noqa: A002
                mode='MIN_COMBINED', name=None):
    return apply_op('QuantizeV2',
                    convert_to_tensor(input, dtype=dtypes.float32),
                    convert_to_tensor(float(min_range)),
                    convert_to_tensor(float(max_range)), mode=mode,
                    name=name)


def dequantize(input, min_range, max_range, mode='MIN_COMBINED',  # noqa: A002
               name=None):
    return apply_op('Dequantize', convert_to_tensor(input),
                    convert_to_tensor(min_range),
                    convert_to_tensor(max_range), mode=mode, name=name)


def quantized_matmul(a, b, min_a, max_a, min_b, max_b, transpose_a=False,
                     transpose_b=False, name=None):
    return apply_op('QuantizedMatMul', a, b, convert_to_tensor(min_a),
                    convert_to_tensor(max_a), convert_to_tensor(min_b),
                    convert_to_tensor(max_b), transpose_a=transpose_a,
                    transpose_b=transpose_b, name=name)


def quantized_relu(features, min_features, max_features, name=None):
    return apply_op('QuantizedRelu', features,
                    convert_to_tensor(min_features),
                    convert_to_tensor(max_features), name=name)


def quantize_down_and_shrink_range(input, input_min, input_max,  # noqa: A002
                                   name=None):
    return apply_op('QuantizeDownAndShrinkRange', input,
                    convert_to_tensor(input_min),
                    convert_to_tensor(input_max), name=name)


def requantization_range(input, input_min, input_max, name=None):  # noqa: A002
    return apply_op('RequantizationRange', input,
                    convert_to_tensor(input_min),
                    convert_to_tensor(input_max), name=name)


for _op in ('QuantizeV2', 'Dequantize', 'QuantizedMatMul', 'QuantizedRelu',
            'QuantizeDownAndShrinkRange', 'RequantizationRange'):
    NoGradient(_op)
