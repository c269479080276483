"""graph_transforms CLI (reference tools/graph_transforms/
transform_graph_main.cc — the transform registry is python here, over the
same GraphDef wire format).

Usage: python -m simple_tensorflow_amd.python.tools.transform_graph \
  --in_graph g.pb --out_graph out.pb --inputs x --outputs y \
  --transforms 'strip_unused_nodes fold_batch_norms remove_nodes(op=Identity)'
"""
import argparse
import re

from simple_tensorflow_amd.python.framework import dtypes, graph_util


def _parse_transform(spec):
    m = re.match(r'(\w+)(?:\((.*)\))?$', spec)
    if not m:
        raise ValueError('bad transform spec %r' % spec)
    name = m.group(1)
    params = {}
    if m.group(2):
        for kv in m.group(2).split(','):
            k, _, v = kv.partition('=')
            params.setdefault(k.strip(), []).append(v.strip())
    return name, params


def strip_unused_nodes(gd, inputs, outputs, params):
    from simple_tensorflow_amd.python.tools import strip_unused_lib
    if not inputs:
        return graph_util.extract_sub_graph(gd, outputs)
    return strip_unused_lib.strip_unused(gd, inputs, outputs,
                                         int(dtypes.float32))


def remove_nodes(gd, inputs, outputs, params):
    ops_to_remove = params.get('op', ['Identity', 'CheckNumerics'])
    nodes = graph_util._as_nodes(gd)
    protected = set(inputs) | set(outputs)
    # reuse remove_training_nodes' rewiring for the pass-through set
    removable_ops = set(ops_to_remove)
    out = graph_util.remove_training_nodes(
        graph_util._serialize(
            [n for n in nodes]), protected_nodes=list(protected))
    # remove_training_nodes only handles Identity/CheckNumerics; for other
    # single-input ops do a generic pass
    nodes = graph_util._as_nodes(out)
    removable = {n['name']: n['input'][0] for n in nodes
                 if n['op'] in removable_ops and n['name'] not in protected
                 and len(n['input']) == 1 and
                 not n['input'][0].startswith('^')}
    result = []
    for n in nodes:
        if n['name'] in removable:
            continue
        n = dict(n)
        fixed = []
        for i in n['input']:
            base = graph_util._base_name(i)
            seen = set()
            while base in removable and base not in seen:
                seen.add(base)
                base = graph_util._base_name(removable[base])
            fixed.append(('^' + base) if i.startswith('^') else base)
        n['input'] = fixed
        result.append(n)
    return graph_util._serialize(result)


def fold_batch_norms(gd, inputs, outputs, params):
    from simple_tensorflow_amd.python.tools import optimize_for_inference_lib
    return optimize_for_inference_lib.fold_batch_norms(gd)


def sort_by_execution_order(gd, inputs, outputs, params):
    nodes = graph_util._as_nodes(gd)
    by_name = {n['name']: n for n in nodes}
    order, seen = [], set()

    def visit(name):
        if name in seen or name not in by_name:
            return
        seen.add(name)
        for i in by_name[name]['input']:
            visit(graph_util._base_name(i))
        order.append(by_name[name])

    for n in nodes:
        visit(n['name'])
    return graph_util._serialize(order)


TRANSFORMS = {
    'strip_unused_nodes': strip_unused_nodes,
    'remove_nodes': remove_nodes,
    'fold_batch_norms': fold_batch_norms,
    'sort_by_execution_order': sort_by_execution_order,
}


def transform_graph(gd, inputs, outputs, transforms):
    for spec in transforms:
        name, params = _parse_transform(spec)
        fn = TRANSFORMS.get(name)
        if fn is None:
            raise ValueError('unknown transform %r (have: %s)' %
                             (name, ', '.join(sorted(TRANSFORMS))))
        gd = fn(gd, inputs, outputs, params)
    return gd


def main():
    p = argparse.ArgumentParser()
    p.add_argument('--in_graph', required=True)
    p.add_argument('--out_graph', required=True)
    p.add_argument('--inputs', default='')
    p.add_argument('--outputs', required=True)
    p.add_argument('--transforms', required=True)
    a = p.parse_args()
    with open(a.in_graph, 'rb') as f:
        gd = f.read()
    inputs = [x for x in a.inputs.split(',') if x]
    outputs = [x for x in a.outputs.split(',') if x]
    specs = re.findall(r'\w+(?:\([^)]*\))?', a.transforms)
    gd = transform_graph(gd, inputs, outputs, specs)
    with open(a.out_graph, 'wb') as f:
        f.write(gd)


if __name__ == '__main__':
    main()
