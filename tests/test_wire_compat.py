"""Protobuf wire-format compatibility evidence.

pbwire/pbreader round-tripping each other proves self-consistency, not
format correctness. These tests break that circularity two ways:
1. fixtures: GraphDef/TensorProto byte strings hand-assembled in this file
   directly from the protobuf wire spec + tensorflow's .proto field numbers
   (graph.proto: node=1; node_def.proto: name=1 op=2 input=3 device=4
   attr=5; attr_value.proto: s=2 i=3 f=4 b=5 type=6 shape=7 tensor=8;
   tensor.proto: dtype=1 tensor_shape=2 tensor_content=4) — our readers
   must parse them.
2. an independent minimal varint/TLV decoder written here (no pbwire
   imports) parses what our writers emit.
"""
import struct

import numpy as np

import simple_tensorflow_amd as tf
from simple_tensorflow_amd.python.framework import pbreader, pbwire


# ---- tiny independent encoder (wire spec only) ----
def _vint(n):
    out = b''
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out += bytes([b | 0x80])
        else:
            out += bytes([b])
            return out


def _tag(field, wire):
    return _vint((field << 3) | wire)


def _ld(field, payload):  # length-delimited
    return _tag(field, 2) + _vint(len(payload)) + payload


def _vi(field, value):
    return _tag(field, 0) + _vint(value)


def test_hand_assembled_graphdef_parses():
    # GraphDef { node { name:"x" op:"Const"
    #   attr { key:"dtype" value { type: DT_FLOAT } }
    #   attr { key:"value" value { tensor {
    #     dtype: DT_FLOAT
    #     tensor_shape { dim { size: 2 } }
    #     tensor_content: <2 f32 LE> } } } }
    #   node { name:"y" op:"Identity" input:"x" } }
    content = struct.pack('<2f', 1.5, -2.5)
    tshape = _ld(2, _vi(1, 2))  # TensorShapeProto.dim[0].size = 2
    tproto = _vi(1, 1) + _ld(2, tshape) + _ld(4, content)
    attr_dtype = _ld(1, b'dtype') + _ld(2, _vi(6, 1))
    attr_value = _ld(1, b'value') + _ld(2, _ld(8, tproto))
    node_x = _ld(1, b'x') + _ld(2, b'Const') + _ld(5, attr_dtype) + \
        _ld(5, attr_value)
    node_y = _ld(1, b'y') + _ld(2, b'Identity') + _ld(3, b'x')
    gdef = _ld(1, node_x) + _ld(1, node_y)

    nodes = pbreader.parse_graph_def(gdef)
    assert [n['name'] for n in nodes] == ['x', 'y']
    assert nodes[0]['op'] == 'Const'
    assert nodes[1]['input'] == ['x']
    kind, dt = nodes[0]['attr']['dtype']
    assert kind == 'type' and int(dt) == 1
    kind, traw = nodes[0]['attr']['value']
    assert kind == 'tensor'
    arr = pbreader.np_from_tensor_proto(traw)
    np.testing.assert_allclose(arr, [1.5, -2.5])


def test_hand_assembled_graphdef_imports_and_runs():
    content = struct.pack('<3f', 1.0, 2.0, 3.0)
    tshape = _ld(2, _vi(1, 3))
    tproto = _vi(1, 1) + _ld(2, tshape) + _ld(4, content)
    attr_dtype = _ld(1, b'dtype') + _ld(2, _vi(6, 1))
    attr_value = _ld(1, b'value') + _ld(2, _ld(8, tproto))
    node = _ld(1, b'fixture_const') + _ld(2, b'Const') + \
        _ld(5, attr_dtype) + _ld(5, attr_value)
    gdef = _ld(1, node)

    tf.reset_default_graph()
    from simple_tensorflow_amd.python.framework import importer
    out = importer.import_graph_def(gdef,
                                    return_elements=['fixture_const:0'],
                                    name='fx')
    with tf.Session() as s:
        np.testing.assert_allclose(s.run(out[0]), [1.0, 2.0, 3.0])


# ---- independent minimal decoder ----
def _decode_fields(data):
    """Raw TLV decode with no pbreader involvement."""
    out = []
    off = 0
    while off < len(data):
        tag = 0
        shift = 0
        while True:
            b = data[off]
            off += 1
            tag |= (b & 0x7F) << shift
            if not b & 0x80:
                break
            shift += 7
        field, wire = tag >> 3, tag & 7
        if wire == 0:
            val = 0
            shift = 0
            while True:
                b = data[off]
                off += 1
                val |= (b & 0x7F) << shift
                if not b & 0x80:
                    break
                shift += 7
            out.append((field, val))
        elif wire == 2:
            ln = 0
            shift = 0
            while True:
                b = data[off]
                off += 1
                ln |= (b & 0x7F) << shift
                if not b & 0x80:
                    break
                shift += 7
            out.append((field, data[off:off + ln]))
            off += ln
        elif wire == 5:
            out.append((field, data[off:off + 4]))
            off += 4
        elif wire == 1:
            out.append((field, data[off:off + 8]))
            off += 8
        else:
            raise ValueError('wire type %d' % wire)
    return out


def test_our_graphdef_decodes_independently():
    tf.reset_default_graph()
    a = tf.constant(np.array([4.0, 5.0], np.float32), name='aa')
    b = tf.identity(a, name='bb')
    gd = tf.get_default_graph().as_graph_def()
    fields = _decode_fields(bytes(gd))
    nodes = [v for f, v in fields if f == 1]
    assert len(nodes) == 2
    n0 = dict()
    for f, v in _decode_fields(nodes[0]):
        n0.setdefault(f, []).append(v)
    assert n0[1] == [b'aa'] and n0[2] == [b'Const']
    n1 = dict()
    for f, v in _decode_fields(nodes[1]):
        n1.setdefault(f, []).append(v)
    assert n1[1] == [b'bb'] and n1[2] == [b'Identity'] and n1[3] == [b'aa']
    # dig out the tensor content of the const's value attr
    attrs = {}
    for raw in n0.get(5, []):
        kv = dict((f, v) for f, v in _decode_fields(raw))
        attrs[kv[1].decode()] = kv[2]
    tensor_field = dict(
        (f, v) for f, v in _decode_fields(attrs['value']))[8]
    tfields = {}
    for f, v in _decode_fields(tensor_field):
        tfields.setdefault(f, []).append(v)
    assert tfields[1] == [1]  # DT_FLOAT
    np.testing.assert_allclose(
        np.frombuffer(tfields[4][0], np.float32), [4.0, 5.0])


def test_tfevents_record_framing():
    """TFRecord framing of an events file: length(8) + masked crc(4) +
    payload + crc(4); verify the length header independently."""
    import os
    import tempfile
    from simple_tensorflow_amd.python.summary import summary as summ
    d = tempfile.mkdtemp()
    w = tf.summary.FileWriter(d)
    w.close()
    files = [f for f in os.listdir(d) if 'tfevents' in f]
    assert files
    data = open(os.path.join(d, files[0]), 'rb').read()
    ln = struct.unpack('<Q', data[:8])[0]
    assert 0 < ln < len(data)
    payload = data[12:12 + ln]
    # Event proto: field 1 = wall_time (double, wire 1)
    fields = _decode_fields(payload)
    assert any(f == 1 for f, _ in fields)
