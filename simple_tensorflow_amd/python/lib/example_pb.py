"""tf.train.Example wire format (example.proto / feature.proto): builders +
parser over the protobuf wire spec — the serialized bytes interoperate with
any stock TFRecord tooling.

Example { features: Features = 1 }
Features { feature: map<string, Feature> = 1 }
Feature  { bytes_list = 1 | float_list = 2 | int64_list = 3 }
BytesList { value: repeated bytes = 1 }
FloatList { value: repeated float = 1 (packed) }
Int64List { value: repeated int64 = 1 (packed) }
"""
import struct

import numpy as np

from simple_tensorflow_amd.python.framework import pbwire
from simple_tensorflow_amd.python.framework.pbreader import _fields


class BytesList(object):
    def __init__(self, value=None):
        self.value = [v.encode() if isinstance(v, str) else bytes(v)
                      for v in (value or [])]


class FloatList(object):
    def __init__(self, value=None):
        self.value = [float(v) for v in (value or [])]


class Int64List(object):
    def __init__(self, value=None):
        self.value = [int(v) for v in (value or [])]


class Feature(object):
    def __init__(self, bytes_list=None, float_list=None, int64_list=None):
        self.bytes_list = bytes_list
        self.float_list = float_list
        self.int64_list = int64_list

    def _serialize(self):
        if self.bytes_list is not None:
            body = b''.join(pbwire.f_bytes(1, v)
                            for v in self.bytes_list.value)
            return pbwire.f_bytes(1, body)
        if self.float_list is not None:
            packed = struct.pack('<%df' % len(self.float_list.value),
                                 *self.float_list.value)
            return pbwire.f_bytes(2, pbwire.f_bytes(1, packed))
        if self.int64_list is not None:
            body = b''.join(pbwire.varint(v if v >= 0 else v + (1 << 64))
                            for v in self.int64_list.value)
            return pbwire.f_bytes(3, pbwire.f_bytes(1, body))
        return b''


class Features(object):
    def __init__(self, feature=None):
        self.feature = dict(feature or {})

    def _serialize(self):
        out = b''
        for k in sorted(self.feature):
            entry = pbwire.f_bytes(1, k) + \
                pbwire.f_bytes(2, self.feature[k]._serialize())
            out += pbwire.f_bytes(1, entry)
        return out


class Example(object):
    def __init__(self, features=None):
        self.features = features or Features()

    def SerializeToString(self):
        return pbwire.f_bytes(1, self.features._serialize())


def _read_varint(data, off=0):
    val = 0
    shift = 0
    while True:
        b = data[off]
        off += 1
        val |= (b & 0x7F) << shift
        if not b & 0x80:
            return val, off
        shift += 7


def parse_example_bytes(blob):
    """bytes -> {name: list-of-bytes | np.float32 array | np.int64 array}."""
    out = {}
    for f, w, v in _fields(bytes(blob)):
        if f != 1:
            continue
        for f2, w2, entry in _fields(v):
            if f2 != 1:
                continue
            key = None
            feat = None
            for f3, w3, v3 in _fields(entry):
                if f3 == 1:
                    key = v3.decode()
                elif f3 == 2:
                    feat = v3
            if key is None or feat is None:
                continue
            for f4, w4, v4 in _fields(feat):
                if f4 == 1:  # bytes list
                    out[key] = [bv for f5, _, bv in _fields(v4) if f5 == 1]
                elif f4 == 2:  # float list (packed or repeated)
                    vals = []
                    for f5, w5, v5 in _fields(v4):
                        if f5 != 1:
                            continue
                        if w5 == 2:
                            vals.extend(struct.unpack(
                                '<%df' % (len(v5) // 4), v5))
                        else:
                            vals.append(struct.unpack('<f', v5)[0])
                    out[key] = np.array(vals, np.float32)
                elif f4 == 3:  # int64 list
                    vals = []
                    for f5, w5, v5 in _fields(v4):
                        if f5 != 1:
                            continue
                        if w5 == 2:
                            off = 0
                            while off < len(v5):
                                x, off = _read_varint(v5, off)
                                if x >= 1 << 63:
                                    x -= 1 << 64
                                vals.append(x)
                        else:
                            x = v5
                            if x >= 1 << 63:
                                x -= 1 << 64
                            vals.append(x)
                    out[key] = np.array(vals, np.int64)
    return out
