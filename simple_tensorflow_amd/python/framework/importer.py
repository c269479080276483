"""import_graph_def: rebuild Operations in the default graph from serialized
GraphDef bytes (capability analog of reference
tensorflow/python/framework/importer.py — re-implemented on our wire reader).

Back-edges (Merge <- NextIteration inside while loops) are handled by a
two-pass build: all nodes are created first with empty inputs, then every
input list is patched, so node order in the GraphDef does not matter.
"""
from simple_tensorflow_amd.python.framework import ops
from simple_tensorflow_amd.python.framework import pbreader


def _output_dtypes(od, attrs):
    attr_defs = {a['name']: a for a in od['attr']}
    out_dtypes, out_ref = [], []
    for arg in od['output_arg']:
        n = 1
        if arg['number_attr']:
            if arg['number_attr'] in attrs:
                n = attrs[arg['number_attr']][1]
            else:
                n = ops._attr_default(attr_defs, arg['number_attr'])
        if arg['type_list_attr']:
            lv = attrs[arg['type_list_attr']][1]
            for t in lv['type']:
                out_dtypes.append(int(t))
                out_ref.append(arg['is_ref'])
            continue
        if arg['type_attr']:
            if arg['type_attr'] in attrs:
                dt = attrs[arg['type_attr']][1]
            else:
                dt = ops._attr_default(attr_defs, arg['type_attr'])
        else:
            dt = arg['type']
        for _ in range(int(n)):
            out_dtypes.append(int(dt))
            out_ref.append(arg['is_ref'])
    return out_dtypes, out_ref


def import_graph_def(graph_def, input_map=None, return_elements=None,
                     name=None, op_dict=None, producer_op_list=None):
    """Imports `graph_def` (bytes, or a pre-parsed node list) into the
    default graph under name scope `name` (default 'import').

    input_map maps 'node:port' strings in the GraphDef to existing Tensors;
    return_elements is a list of 'node' / 'node:port' strings to return.
    """
    if isinstance(graph_def, (bytes, bytearray)):
        nodes = pbreader.parse_graph_def(bytes(graph_def))
    else:
        nodes = list(graph_def)
    input_map = dict(input_map or {})
    g = ops.get_default_graph()
    if name is None:
        name = 'import'
    prefix = g.unique_name(name, mark_as_used=True) if name else ''
    reg = ops.op_defs()

    # Pass 1: create every op (inputs patched in pass 2 — back edges).
    created = {}
    for nd in nodes:
        od = reg.get(nd['op'])
        if od is None:
            raise ValueError("No op named '%s' in the registry (node %s)" %
                             (nd['op'], nd['name']))
        attrs = dict(nd['attr'])
        out_dtypes, out_ref = _output_dtypes(od, attrs)
        full = (prefix + '/' + nd['name']) if prefix else nd['name']
        if full in g._nodes_by_name:
            raise ValueError("Duplicate node name '%s' in import" % full)
        op = ops.Operation(g, full, nd['op'], [], [], attrs,
                           nd.get('device', ''), out_dtypes, out_ref)
        with g._lock:
            g._nodes_by_name[full] = op
            g._node_list.append(op)
            g._names_used[full] = 1
            # Mark ancestor scopes used so later name_scope()s uniquify
            # around the imported names (save/* -> save_1/*).
            parts = full.split('/')
            for i in range(1, len(parts)):
                g._names_used.setdefault('/'.join(parts[:i]), 1)
            g.version += 1
        created[nd['name']] = op

    # Pass 2: wire inputs.
    def resolve(inp):
        base, _, idx = inp.partition(':')
        idx = int(idx or 0)
        key = '%s:%d' % (base, idx)
        if key in input_map:
            return input_map[key]
        if idx == 0 and base in input_map:
            return input_map[base]
        src = created.get(base)
        if src is None:
            raise ValueError("Input node '%s' not found in GraphDef" % base)
        return src.outputs[idx]

    for nd in nodes:
        op = created[nd['name']]
        for inp in nd['input']:
            if inp.startswith('^'):
                cname = inp[1:]
                if cname in created:
                    op.control_inputs.append(created[cname])
                elif cname in input_map:
                    op.control_inputs.append(input_map[cname].op)
                else:
                    raise ValueError(
                        "Control input '%s' not found in GraphDef" % cname)
            else:
                op.inputs.append(resolve(inp))

    # Best-effort shape inference in GraphDef order (back edges tolerated).
    for nd in nodes:
        try:
            ops._infer_shapes(created[nd['name']])
        except Exception:
            pass

    if return_elements is None:
        return None
    out = []
    for el in return_elements:
        if ':' in el:
            base, _, idx = el.partition(':')
            out.append(created[base].outputs[int(idx)])
        else:
            out.append(created[el])
    return out
