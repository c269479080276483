"""GraphDef-level surgery: extract_sub_graph, convert_variables_to_constants,
remove_training_nodes (capability analog of reference
python/framework/graph_util_impl.py; used by tools/freeze_graph)."""
import numpy as np

from simple_tensorflow_amd.python.framework import dtypes, pbreader, pbwire


def _as_nodes(graph_def):
    if isinstance(graph_def, (bytes, bytearray)):
        return pbreader.parse_graph_def(bytes(graph_def))
    return [dict(n) for n in graph_def]


def _serialize(nodes):
    return pbwire.graph_def([
        pbwire.node_def(n['name'], n['op'], n['input'],
                        n.get('device', ''), n['attr']) for n in nodes])


def _base_name(inp):
    if inp.startswith('^'):
        inp = inp[1:]
    return inp.split(':')[0]


def extract_sub_graph(graph_def, dest_nodes):
    """Keep only nodes reachable (via data or control inputs) from
    dest_nodes."""
    nodes = _as_nodes(graph_def)
    by_name = {n['name']: n for n in nodes}
    keep = set()
    stack = list(dest_nodes)
    while stack:
        name = _base_name(stack.pop())
        if name in keep:
            continue
        if name not in by_name:
            raise ValueError("node '%s' not in graph" % name)
        keep.add(name)
        stack.extend(by_name[name]['input'])
    out = [n for n in nodes if n['name'] in keep]
    return _serialize(out)


def must_run_on_cpu(node, pin_variables_on_cpu=False):
    return node['op'] in ('Variable', 'VariableV2') and pin_variables_on_cpu


def convert_variables_to_constants(sess, input_graph_def, output_node_names,
                                   variable_names_whitelist=None):
    """Replaces VariableV2 nodes with Const nodes holding their current
    session value, then prunes to what output_node_names reach."""
    nodes = _as_nodes(input_graph_def)
    var_names = [n['name'] for n in nodes
                 if n['op'] in ('Variable', 'VariableV2')
                 and (variable_names_whitelist is None
                      or n['name'] in variable_names_whitelist)]
    values = sess.run([v + ':0' for v in var_names]) if var_names else []
    by_name = dict(zip(var_names, values))
    out = []
    converted = 0
    for n in nodes:
        if n['name'] in by_name:
            val = np.asarray(by_name[n['name']])
            dt = dtypes.as_dtype(n['attr']['dtype'][1]) if 'dtype' in \
                n['attr'] else dtypes.as_dtype(val.dtype)
            content = np.ascontiguousarray(
                val.astype(dt.as_numpy_dtype)).tobytes()
            tp = pbwire.tensor_proto(int(dt), list(val.shape),
                                     content=content)
            out.append({'name': n['name'], 'op': 'Const', 'input': [],
                        'device': n.get('device', ''),
                        'attr': {'dtype': ('type', int(dt)),
                                 'value': ('tensor', tp)}})
            converted += 1
        elif n['op'] == 'Assign' and _base_name(n['input'][0]) in by_name:
            continue  # initializer of a frozen variable
        else:
            out.append(n)
    return extract_sub_graph(out, output_node_names)


def remove_training_nodes(input_graph_def, protected_nodes=None):
    """Drops Identity/CheckNumerics pass-through nodes and rewires their
    consumers (reference graph_util_impl.remove_training_nodes)."""
    protected = set(protected_nodes or [])
    nodes = _as_nodes(input_graph_def)
    removable = {}
    for n in nodes:
        if n['op'] in ('Identity', 'CheckNumerics') and \
                n['name'] not in protected and len(n['input']) == 1 and \
                not n['input'][0].startswith('^'):
            removable[n['name']] = n['input'][0]

    def resolve(inp):
        ctrl = inp.startswith('^')
        base = inp[1:] if ctrl else inp
        port = ''
        if ':' in base:
            base, _, port = base.partition(':')
        seen = set()
        while base in removable and base not in seen:
            seen.add(base)
            nxt = removable[base]
            base, _, port2 = nxt.partition(':')
            port = port2 or '0'
        full = base + ((':' + port) if port and port != '0' else '')
        return ('^' + base) if ctrl else full

    out = []
    for n in nodes:
        if n['name'] in removable:
            continue
        n = dict(n)
        n['input'] = [resolve(i) for i in n['input']]
        out.append(n)
    return _serialize(out)
