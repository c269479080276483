"""MetaGraphDef export/import — byte-compatible wire layout with the
reference core/protobuf/meta_graph.proto (meta_info_def=1, graph_def=2,
saver_def=3, collection_def=4) and framework/variable.proto VariableDef
(variable_name=1, initializer_name=2, snapshot_name=3).

Analog of reference python/framework/meta_graph.py +
python/training/saver.py export_meta_graph/import_meta_graph."""
from simple_tensorflow_amd.python.framework import ops
from simple_tensorflow_amd.python.framework import pbreader
from simple_tensorflow_amd.python.framework.pbwire import (
    f_bytes, f_varint, f_float)

# Collections whose elements are Variables: stored as bytes_list of
# VariableDef (like the reference); everything else that holds
# Tensors/Operations is stored as a node_list of names.
_VARIABLE_COLLECTIONS = frozenset([
    'variables', 'trainable_variables', 'local_variables',
    'moving_average_variables', 'model_variables'])


def _variable_def(v):
    return (f_bytes(1, v._variable.name) +
            f_bytes(2, v._initializer_op.name if hasattr(
                v._initializer_op, 'name') else v._initializer_op.name) +
            f_bytes(3, v._snapshot.name))


def _element_name(x):
    if hasattr(x, 'name'):
        return x.name
    return str(x)


def _collection_def(name, items):
    if not items:
        return None
    from simple_tensorflow_amd.python.ops import variables as var_mod
    if name in _VARIABLE_COLLECTIONS and isinstance(items[0],
                                                    var_mod.Variable):
        body = b''
        for v in items:
            body += f_bytes(1, _variable_def(v))
        return f_bytes(2, body)  # bytes_list = 2
    try:
        names = [_element_name(x) for x in items]
    except Exception:
        return None
    body = b''
    for n in names:
        body += f_bytes(1, n)
    return f_bytes(1, body)  # node_list = 1


def saver_def_bytes(saver):
    """SaverDef: filename_tensor_name=1, save_tensor_name=2,
    restore_op_name=3, max_to_keep=4, version=7 (V2)."""
    return (f_bytes(1, saver._filename.name) +
            f_bytes(2, saver._save_op.name if hasattr(saver._save_op, 'name')
                    else str(saver._save_op)) +
            f_bytes(3, saver._restore_op.name) +
            f_varint(4, saver._max_to_keep) +
            f_varint(7, 2))


def export_meta_graph(filename=None, graph=None, saver=None,
                      collection_list=None, meta_info=b''):
    """Serialize graph + collections (+ optional SaverDef) to MetaGraphDef
    bytes; write to `filename` if given."""
    g = graph or ops.get_default_graph()
    out = b''
    if meta_info:
        out += f_bytes(1, meta_info)
    out += f_bytes(2, g.as_graph_def())
    if saver is not None:
        out += f_bytes(3, saver_def_bytes(saver))
    names = collection_list if collection_list is not None else list(
        g._collections)
    for cname in names:
        cd = _collection_def(cname, g.get_collection(cname))
        if cd is None:
            continue
        entry = f_bytes(1, cname) + f_bytes(2, cd)
        out += f_bytes(4, entry)
    if filename:
        with open(filename, 'wb') as f:
            f.write(out)
    return out


def _parse_collection_def(data):
    for f, w, v in pbreader._fields(data):
        if f == 1:  # node_list
            names = []
            for f2, _, v2 in pbreader._fields(v):
                if f2 == 1:
                    names.append(v2.decode())
            return ('node_list', names)
        if f == 2:  # bytes_list
            vals = []
            for f2, _, v2 in pbreader._fields(v):
                if f2 == 1:
                    vals.append(bytes(v2))
            return ('bytes_list', vals)
    return ('node_list', [])


def _parse_variable_def(data):
    d = {1: '', 2: '', 3: ''}
    for f, w, v in pbreader._fields(data):
        if f in d:
            d[f] = v.decode()
    return d[1], d[2], d[3]


def parse_meta_graph(data):
    gd = None
    saver_def = {}
    collections = {}
    for f, w, v in pbreader._fields(data):
        if f == 2:
            gd = bytes(v)
        elif f == 3:
            for f2, _, v2 in pbreader._fields(v):
                if f2 in (1, 2, 3):
                    saver_def[f2] = v2.decode()
                elif f2 == 4:
                    saver_def['max_to_keep'] = v2
        elif f == 4:
            key, cd = None, None
            for f2, _, v2 in pbreader._fields(v):
                if f2 == 1:
                    key = v2.decode()
                elif f2 == 2:
                    cd = _parse_collection_def(v2)
            if key is not None and cd is not None:
                collections[key] = cd
    return gd, saver_def, collections


def import_meta_graph(meta_graph_or_file, clear_devices=False,
                      import_scope=None):
    """Rebuilds the graph (+ variable collections) in the default graph and
    returns a Saver wired to the imported save/restore ops (or None)."""
    if isinstance(meta_graph_or_file, str):
        with open(meta_graph_or_file, 'rb') as f:
            data = f.read()
    else:
        data = bytes(meta_graph_or_file)
    gd, saver_def, collections = parse_meta_graph(data)
    if gd is None:
        raise ValueError('MetaGraphDef has no graph_def')
    nodes = pbreader.parse_graph_def(gd)
    if clear_devices:
        for n in nodes:
            n['device'] = ''
    from simple_tensorflow_amd.python.framework import importer
    importer.import_graph_def(nodes, name=import_scope or '')
    g = ops.get_default_graph()
    prefix = (import_scope + '/') if import_scope else ''

    from simple_tensorflow_amd.python.ops import variables as var_mod
    for cname, (kind, vals) in collections.items():
        if kind == 'bytes_list' and cname in _VARIABLE_COLLECTIONS:
            for vd in vals:
                vname, iname, sname = _parse_variable_def(vd)
                v = var_mod.Variable._from_graph_elements(
                    g.get_tensor_by_name(prefix + vname),
                    g.get_operation_by_name(
                        prefix + iname.split(':')[0]),
                    g.get_tensor_by_name(prefix + sname))
                g.add_to_collection(cname, v)
        elif kind == 'node_list':
            for n in vals:
                try:
                    if ':' in n:
                        g.add_to_collection(cname,
                                            g.get_tensor_by_name(prefix + n))
                    else:
                        g.add_to_collection(
                            cname, g.get_operation_by_name(prefix + n))
                except KeyError:
                    pass

    if saver_def:
        from simple_tensorflow_amd.python.training import saver as saver_mod
        return saver_mod.Saver._from_imported(
            g.get_tensor_by_name(prefix + saver_def[1]),
            g.get_operation_by_name(prefix + saver_def[2].split(':')[0]),
            g.get_operation_by_name(prefix + saver_def[3]),
            max_to_keep=int(saver_def.get('max_to_keep', 5)))
    return None
