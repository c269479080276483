"""optimize_for_inference: strip training-only nodes, fold frozen batch
norms into the preceding conv's weights (capability analog of reference
python/tools/optimize_for_inference_lib.py:70,175)."""
import numpy as np

from simple_tensorflow_amd.python.framework import (dtypes, graph_util,
                                                    pbreader, pbwire)


def _const_value(node):
    if node['op'] != 'Const':
        return None
    return pbreader.np_from_tensor_proto(node['attr']['value'][1])


def _make_const(name, arr, dtype_enum):
    npdt = dtypes.as_dtype(dtype_enum).as_numpy_dtype
    content = np.ascontiguousarray(arr.astype(npdt)).tobytes()
    tp = pbwire.tensor_proto(dtype_enum, list(arr.shape), content=content)
    return {'name': name, 'op': 'Const', 'input': [], 'device': '',
            'attr': {'dtype': ('type', dtype_enum),
                     'value': ('tensor', tp)}}


def fold_batch_norms(input_graph_def):
    """Folds BatchNormMi / FusedBatchNorm (inference form, all-Const
    scale/offset/mean/var after freezing) into the upstream Conv2D's Const
    filter: conv(x, w') + b' with w' = w*gamma/sqrt(var+eps) and
    b' = beta - mean*gamma/sqrt(var+eps)."""
    nodes = graph_util._as_nodes(input_graph_def)
    by_name = {n['name']: n for n in nodes}
    removed = set()
    rewrites = {}  # old tensor name -> new tensor name
    new_nodes = []

    for n in nodes:
        if n['op'] not in ('FusedBatchNorm', 'BatchNormMi'):
            continue
        src = by_name.get(graph_util._base_name(n['input'][0]))
        if src is None or src['op'] != 'Conv2D':
            continue
        w_node = by_name.get(graph_util._base_name(src['input'][1]))
        params = [by_name.get(graph_util._base_name(i))
                  for i in n['input'][1:5]]
        if w_node is None or w_node['op'] != 'Const' or \
                any(p is None or p['op'] != 'Const' for p in params):
            continue
        w = _const_value(w_node).astype(np.float32)
        gamma, beta, mean, var = [
            _const_value(p).astype(np.float32).reshape(-1) for p in params]
        eps = 1e-3
        for k in ('epsilon',):
            if k in n['attr']:
                eps = float(n['attr'][k][1])
        inv = gamma / np.sqrt(var + eps)
        w_f = w * inv.reshape(1, 1, 1, -1)
        b_f = beta - mean * inv
        wdt = w_node['attr']['dtype'][1]
        new_nodes.append(_make_const(w_node['name'] + '_bnfold', w_f, wdt))
        new_nodes.append(_make_const(n['name'] + '_bias', b_f,
                                     int(dtypes.float32)))
        # Rewire: conv reads folded weights; a BiasAdd replaces the BN under
        # the BN's own name, so downstream consumers keep working.
        src['input'][1] = w_node['name'] + '_bnfold'
        new_nodes.append({
            'name': n['name'], 'op': 'BiasAdd',
            'input': [n['input'][0], n['name'] + '_bias'],
            'device': n.get('device', ''),
            'attr': {'T': dict(n['attr']).get('T', ('type',
                                                    int(dtypes.float32)))}})
        removed.add(n['name'])

    out = []
    for n in nodes:
        if n['name'] in removed:
            continue
        n = dict(n)
        fixed = []
        for i in n['input']:
            ctrl = i.startswith('^')
            base = graph_util._base_name(i)
            port = i.split(':')[1] if ':' in i else '0'
            if base in rewrites and (not ctrl) and port == '0':
                fixed.append(rewrites[base])
            else:
                fixed.append(i)
        n['input'] = fixed
        out.append(n)
    out.extend(new_nodes)
    return graph_util._serialize(out)


def optimize_for_inference(input_graph_def, input_node_names,
                           output_node_names, placeholder_type_enum):
    from simple_tensorflow_amd.python.tools import strip_unused_lib
    gd = graph_util.remove_training_nodes(
        input_graph_def, protected_nodes=list(input_node_names) +
        list(output_node_names))
    gd = fold_batch_norms(gd)
    gd = strip_unused_lib.strip_unused(gd, input_node_names,
                                       output_node_names,
                                       placeholder_type_enum)
    return gd
