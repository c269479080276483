"""Minimal protobuf wire-format writer (mirrors csrc/core/pb.h).

Serializes GraphDef/NodeDef/AttrValue/TensorProto byte-compatibly with the
reference schemas (reference: tensorflow/core/framework/*.proto) without a
protobuf dependency.
"""
import struct


def varint(v):
    out = bytearray()
    v &= (1 << 64) - 1
    while v >= 0x80:
        out.append((v & 0x7F) | 0x80)
        v >>= 7
    out.append(v)
    return bytes(out)


def tag(field, wire):
    return varint((field << 3) | wire)


def f_varint(field, v):
    return tag(field, 0) + varint(v)


def f_bytes(field, b):
    if isinstance(b, str):
        b = b.encode()
    return tag(field, 2) + varint(len(b)) + b


def f_float(field, v):
    return tag(field, 5) + struct.pack('<f', v)


def f_packed_varints(field, vs):
    body = b''.join(varint(v) for v in vs)
    return f_bytes(field, body)


def f_packed_floats(field, vs):
    body = struct.pack('<%df' % len(vs), *vs)
    return f_bytes(field, body)


def tensor_shape_proto(dims):
    """TensorShapeProto: dim=2 {size=1}, unknown_rank=3."""
    out = b''
    if dims is None:
        return f_varint(3, 1)
    for d in dims:
        out += f_bytes(2, f_varint(1, d if d is not None else -1))
    return out


def tensor_proto(dtype_enum, dims, content=b'', float_vals=None, int_vals=None,
                 int64_vals=None, bool_vals=None, string_vals=None,
                 double_vals=None, half_vals=None):
    """TensorProto: dtype=1, shape=2, content=4, float_val=5, double_val=6,
    int_val=7, string_val=8, int64_val=10, bool_val=11, half_val=13."""
    out = f_varint(1, dtype_enum)
    out += f_bytes(2, tensor_shape_proto(dims))
    if content:
        out += f_bytes(4, content)
    if float_vals:
        out += f_packed_floats(5, float_vals)
    if double_vals:
        out += f_bytes(6, struct.pack('<%dd' % len(double_vals), *double_vals))
    if int_vals:
        out += f_packed_varints(7, int_vals)
    if string_vals:
        for s in string_vals:
            out += f_bytes(8, s)
    if int64_vals:
        out += f_packed_varints(10, int64_vals)
    if bool_vals:
        out += f_packed_varints(11, [1 if b else 0 for b in bool_vals])
    if half_vals:
        out += f_packed_varints(13, half_vals)
    return out


def attr_value(av):
    """av: ('s'|'i'|'f'|'b'|'type'|'shape'|'tensor'|'list', value).

    AttrValue fields: list=1, s=2, i=3, f=4, b=5, type=6, shape=7, tensor=8.
    For 'tensor' the value must already be serialized TensorProto bytes; for
    'shape' a list of dims (or None).
    """
    kind, v = av
    if kind == 's':
        return f_bytes(2, v)
    if kind == 'i':
        return f_varint(3, int(v))
    if kind == 'f':
        return f_float(4, float(v))
    if kind == 'b':
        return f_varint(5, 1 if v else 0)
    if kind == 'type':
        return f_varint(6, int(v))
    if kind == 'shape':
        return f_bytes(7, tensor_shape_proto(v))
    if kind == 'tensor':
        return f_bytes(8, v)
    if kind == 'list':
        # v: dict with optional keys s,i,f,b,type,shape
        body = b''
        for s in v.get('s', []):
            body += f_bytes(2, s)
        if v.get('i'):
            body += f_packed_varints(3, [int(x) for x in v['i']])
        if v.get('f'):
            body += f_packed_floats(4, [float(x) for x in v['f']])
        if v.get('b'):
            body += f_packed_varints(5, [1 if x else 0 for x in v['b']])
        if v.get('type'):
            body += f_packed_varints(6, [int(x) for x in v['type']])
        for sh in v.get('shape', []):
            body += f_bytes(7, tensor_shape_proto(sh))
        return f_bytes(1, body)
    raise ValueError('bad attr kind %r' % kind)


def node_def(name, op, inputs, device, attrs):
    """NodeDef: name=1, op=2, input=3, device=4, attr=5 (map)."""
    out = f_bytes(1, name) + f_bytes(2, op)
    for i in inputs:
        out += f_bytes(3, i)
    if device:
        out += f_bytes(4, device)
    for k in sorted(attrs):
        entry = f_bytes(1, k) + f_bytes(2, attr_value(attrs[k]))
        out += f_bytes(5, entry)
    return out


def graph_def(node_bytes_list, producer=21, library=None):
    """GraphDef: node=1, library=2 (FunctionDefLibrary), versions=4
    {producer=1}."""
    out = b''.join(f_bytes(1, nb) for nb in node_bytes_list)
    if library:
        out += f_bytes(2, library)
    out += f_bytes(4, f_varint(1, producer))
    return out

def arg_def(name, type_enum):
    """OpDef.ArgDef: name=1, type=3."""
    return f_bytes(1, name) + f_varint(3, type_enum)


def op_def_signature(name, input_args, output_args):
    """OpDef subset for FunctionDef.signature: name=1, input_arg=2,
    output_arg=3 (each ArgDef)."""
    out = f_bytes(1, name)
    for n, t in input_args:
        out += f_bytes(2, arg_def(n, t))
    for n, t in output_args:
        out += f_bytes(3, arg_def(n, t))
    return out


def function_def(signature_bytes, node_bytes_list, ret_map):
    """FunctionDef: signature=1 (OpDef), node_def=3 (NodeDef), ret=4
    (map<string,string>)."""
    out = f_bytes(1, signature_bytes)
    for nb in node_bytes_list:
        out += f_bytes(3, nb)
    for k in sorted(ret_map):
        out += f_bytes(4, f_bytes(1, k) + f_bytes(2, ret_map[k]))
    return out


def function_def_library(fdef_bytes_list):
    """FunctionDefLibrary: function=1."""
    return b''.join(f_bytes(1, fb) for fb in fdef_bytes_list)
