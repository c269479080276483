"""Protobuf wire-format reader for GraphDef/NodeDef/AttrValue (the python
mirror of csrc/core/pb.h; used by import_graph_def / meta-graph import)."""
import struct


def _read_varint(data, off):
    v = 0
    shift = 0
    while True:
        b = data[off]
        off += 1
        v |= (b & 0x7F) << shift
        if not (b & 0x80):
            return v, off
        shift += 7


def _fields(data):
    """Yields (field_number, wire_type, value) — value is int for varint,
    bytes for length-delimited, raw 4/8 bytes for fixed."""
    off = 0
    n = len(data)
    while off < n:
        tag, off = _read_varint(data, off)
        field, wire = tag >> 3, tag & 7
        if wire == 0:
            v, off = _read_varint(data, off)
        elif wire == 1:
            v = data[off:off + 8]
            off += 8
        elif wire == 2:
            ln, off = _read_varint(data, off)
            v = data[off:off + ln]
            off += ln
        elif wire == 5:
            v = data[off:off + 4]
            off += 4
        else:
            raise ValueError('bad wire type %d' % wire)
        yield field, wire, v


def parse_tensor_shape(data):
    dims = []
    unknown = False
    for f, w, v in _fields(data):
        if f == 2:
            size = None
            for f2, w2, v2 in _fields(v):
                if f2 == 1:
                    size = v2 if v2 < (1 << 62) else v2 - (1 << 64)
            dims.append(size)
        elif f == 3 and v:
            unknown = True
    return None if unknown else dims


def parse_attr_value(data):
    """Returns our ('kind', value) attr representation."""
    for f, w, v in _fields(data):
        if f == 2:
            return ('s', bytes(v))
        if f == 3:
            return ('i', v if v < (1 << 62) else v - (1 << 64))
        if f == 4:
            return ('f', struct.unpack('<f', v)[0])
        if f == 5:
            return ('b', bool(v))
        if f == 6:
            return ('type', v)
        if f == 7:
            return ('shape', parse_tensor_shape(v))
        if f == 8:
            return ('tensor', bytes(v))
        if f == 1:
            lv = {'s': [], 'i': [], 'f': [], 'b': [], 'type': [], 'shape': []}
            for f2, w2, v2 in _fields(v):
                if f2 == 2:
                    lv['s'].append(bytes(v2))
                elif f2 == 3:
                    if w2 == 2:  # packed
                        off = 0
                        while off < len(v2):
                            x, off = _read_varint(v2, off)
                            lv['i'].append(x)
                    else:
                        lv['i'].append(v2)
                elif f2 == 4:
                    if w2 == 2:
                        for k in range(0, len(v2), 4):
                            lv['f'].append(struct.unpack('<f',
                                                         v2[k:k + 4])[0])
                    else:
                        lv['f'].append(struct.unpack('<f', v2)[0])
                elif f2 == 5:
                    if w2 == 2:
                        for b in v2:
                            lv['b'].append(bool(b & 1))
                    else:
                        lv['b'].append(bool(v2))
                elif f2 == 6:
                    if w2 == 2:
                        off = 0
                        while off < len(v2):
                            x, off = _read_varint(v2, off)
                            lv['type'].append(x)
                    else:
                        lv['type'].append(v2)
                elif f2 == 7:
                    lv['shape'].append(parse_tensor_shape(v2))
            return ('list', lv)
    return ('0', None)


def parse_node_def(data):
    node = {'name': '', 'op': '', 'input': [], 'device': '', 'attr': {}}
    for f, w, v in _fields(data):
        if f == 1:
            node['name'] = v.decode()
        elif f == 2:
            node['op'] = v.decode()
        elif f == 3:
            node['input'].append(v.decode())
        elif f == 4:
            node['device'] = v.decode()
        elif f == 5:
            key = None
            val = ('0', None)
            for f2, w2, v2 in _fields(v):
                if f2 == 1:
                    key = v2.decode()
                elif f2 == 2:
                    val = parse_attr_value(v2)
            if key is not None:
                node['attr'][key] = val
    return node


def parse_graph_def(data):
    nodes = []
    for f, w, v in _fields(data):
        if f == 1:
            nodes.append(parse_node_def(v))
    return nodes


def parse_tensor_proto(data):
    """TensorProto: dtype=1, tensor_shape=2, tensor_content=4, float_val=5,
    int_val=7, string_val=8, int64_val=10, bool_val=11. Returns
    (dtype_enum, dims, content_bytes, repeated_vals)."""
    dtype = 0
    dims = []
    content = b''
    vals = []
    for f, w, v in _fields(data):
        if f == 1:
            dtype = v
        elif f == 2:
            dims = parse_tensor_shape(v) or []
        elif f == 4:
            content = bytes(v)
        elif f == 5:
            if w == 2:
                for k in range(0, len(v), 4):
                    vals.append(struct.unpack('<f', v[k:k + 4])[0])
            else:
                vals.append(struct.unpack('<f', v)[0])
        elif f in (7, 10):
            if w == 2:
                off = 0
                while off < len(v):
                    x, off = _read_varint(v, off)
                    vals.append(x if x < (1 << 62) else x - (1 << 64))
            else:
                vals.append(v if v < (1 << 62) else v - (1 << 64))
        elif f == 8:
            vals.append(bytes(v))
        elif f == 11:
            vals.append(bool(v))
    return dtype, dims, content, vals


def np_from_tensor_proto(data):
    import numpy as np
    from simple_tensorflow_amd.python.framework import dtypes
    dtype, dims, content, vals = parse_tensor_proto(data)
    npdt = dtypes.as_dtype(dtype).as_numpy_dtype
    n = 1
    for d in dims:
        n *= d
    if content:
        arr = np.frombuffer(content, dtype=npdt)
    elif vals:
        arr = np.array(vals, dtype=npdt)
        if arr.size == 1 and n > 1:
            arr = np.full(n, arr[0], dtype=npdt)
    else:
        arr = np.zeros(n, dtype=npdt)
    return arr.reshape(dims)

def parse_arg_def(data):
    arg = {'name': '', 'type': 0}
    for f, w, v in _fields(data):
        if f == 1:
            arg['name'] = v.decode()
        elif f == 3:
            arg['type'] = v
    return arg


def parse_op_def_signature(data):
    sig = {'name': '', 'input_arg': [], 'output_arg': []}
    for f, w, v in _fields(data):
        if f == 1:
            sig['name'] = v.decode()
        elif f == 2:
            sig['input_arg'].append(parse_arg_def(v))
        elif f == 3:
            sig['output_arg'].append(parse_arg_def(v))
    return sig


def parse_function_def(data):
    fdef = {'signature': None, 'node_def': [], 'ret': {}}
    for f, w, v in _fields(data):
        if f == 1:
            fdef['signature'] = parse_op_def_signature(v)
        elif f == 3:
            fdef['node_def'].append(parse_node_def(v))
        elif f == 4:
            k = val = None
            for f2, w2, v2 in _fields(v):
                if f2 == 1:
                    k = v2.decode()
                elif f2 == 2:
                    val = v2.decode()
            if k is not None:
                fdef['ret'][k] = val
    return fdef


def parse_function_def_library(data):
    return [parse_function_def(v) for f, w, v in _fields(data) if f == 1]


def parse_graph_def_full(data):
    """Returns (nodes, functions): the node list plus the FunctionDefLibrary
    content (GraphDef field 2)."""
    nodes, functions = [], []
    for f, w, v in _fields(data):
        if f == 1:
            nodes.append(parse_node_def(v))
        elif f == 2:
            functions.extend(parse_function_def_library(v))
    return nodes, functions
