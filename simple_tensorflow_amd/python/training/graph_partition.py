"""Per-worker graph partitioning for the distributed runtime.

Capability analog of the reference master's BuildAndRegisterPartitions
(master_session.cc:1199 / graph_partition.cc:801 partitioned by WORKER):
the client GraphDef is split by the /job:X/task:N of each node's device;
every cross-worker data edge becomes a publish node on the producer (stores
the tensor in the producer worker's per-step TensorStore) and a recv node on
the consumer (a blocking RecvTensor RPC pull from the producer — the
reference's rpc_rendezvous_mgr.cc:273 pull model). Publish/recv are PyFunc
kernels bound to worker-process callables registered by server_lib.

Edge keys carry the producer address so the consumer knows whom to pull
from: "<producer_addr>|<src_node>:<port>". A per-step int64 'stf_step_id'
placeholder (fed by the master on every RunGraph) scopes the store entries.
"""
import numpy as np

from simple_tensorflow_amd.python.framework import dtypes, ops, pbreader, \
    pbwire

STEP_PH = '__stf_step_id'
PUBLISH_TOKEN = '__dist_publish'
RECV_TOKEN = '__dist_recv'

# dtypes a PyFunc edge can carry natively (numpy-representable)
_F32 = int(dtypes.float32)


def _worker_of(device, default):
    """'/job:ps/task:0/cpu:0' -> ('ps', 0); no job -> default."""
    job, task = None, 0
    for part in device.split('/'):
        if part.startswith('job:'):
            job = part[4:]
        elif part.startswith('task:'):
            task = int(part[5:])
    if job is None:
        return default
    return (job, task)


def _local_device(device):
    """Strip job/task/replica, keep the local device part."""
    keep = [p for p in device.split('/')
            if p and not p.startswith(('job:', 'task:', 'replica:'))]
    return '/' + '/'.join(keep) if keep else ''


def _parse_nodes_raw(graph_def_bytes):
    out = []
    for f, w, v in pbreader._fields(graph_def_bytes):
        if f == 1:
            out.append((v, pbreader.parse_node_def(v)))
    return out


def _edge_dtypes(graph_def_bytes):
    """node name -> (output dtype enums, output is_ref flags), via a
    scratch import."""
    g = ops.Graph()
    with g.as_default():
        from simple_tensorflow_amd.python.framework import importer
        importer.import_graph_def(graph_def_bytes, name='')
    dts = {op.name: [int(t.dtype) for t in op.outputs]
           for op in g._node_list}
    refs = {op.name: [bool(getattr(t, '_is_ref', False))
                      for t in op.outputs]
            for op in g._node_list}
    return dts, refs


def partition_by_worker(graph_def_bytes, cluster, default_worker):
    """Returns (partitions, owner) where partitions maps worker -> GraphDef
    bytes and owner maps node name -> worker. cluster maps (job, task) ->
    address."""
    nodes = _parse_nodes_raw(graph_def_bytes)
    owner = {}
    for raw, nd in nodes:
        owner[nd['name']] = _worker_of(nd['device'], default_worker)
    workers = sorted(set(owner.values()) | {default_worker})
    if len(workers) == 1:
        return {workers[0]: graph_def_bytes}, owner

    out_dtypes, out_refs = _edge_dtypes(graph_def_bytes)

    # Ref edges force colocation (the reference partitioner never splits a
    # ref edge): consumers of a ref output move to the producer's worker,
    # to fixed point (Variable -> Assign -> ... chains).
    changed = True
    while changed:
        changed = False
        for raw, nd in nodes:
            for inp in nd['input']:
                if inp.startswith('^'):
                    continue
                name, _, port_s = inp.partition(':')
                port = int(port_s or 0)
                if out_refs.get(name, [False])[port] and \
                        owner[name] != owner[nd['name']]:
                    owner[nd['name']] = owner[name]
                    changed = True

    parts = {w: [] for w in workers}          # node bytes per worker
    recv_added = {}                           # (edge, worker) -> recv name
    publish_added = set()                     # (src, port) published
    edge_publishes = {}                       # edge str -> (worker, pub name)
    counters = {'n': 0}

    def _key_const(worker, key):
        counters['n'] += 1
        name = '__dist_key_%d' % counters['n']
        tp = pbwire.tensor_proto(int(dtypes.string), [],
                                 string_vals=[key.encode()])
        nb = pbwire.node_def(name, 'Const', [], '',
                             {'dtype': ('type', int(dtypes.string)),
                              'value': ('tensor', tp)})
        parts[worker].append(nb)
        return name

    def _publish(src_worker, src_name, port, dtype_enum, is_control):
        edge = '%s:%d' % (src_name, port) if not is_control else \
            '^' + src_name
        if (src_name, port, is_control) in publish_added:
            return
        publish_added.add((src_name, port, is_control))
        addr = cluster[src_worker]
        key = '%s|%s' % (addr, edge)
        kname = _key_const(src_worker, key)
        counters['n'] += 1
        pname = '__dist_pub_%d' % counters['n']
        if is_control:
            counters['n'] += 1
            dummy = '__dist_dummy_%d' % counters['n']
            tp = pbwire.tensor_proto(int(dtypes.int32), [], int_vals=[0])
            parts[src_worker].append(pbwire.node_def(
                dummy, 'Const', ['^' + src_name], '',
                {'dtype': ('type', int(dtypes.int32)),
                 'value': ('tensor', tp)}))
            data_in = dummy
            tin = [int(dtypes.string), int(dtypes.int64), int(dtypes.int32)]
        else:
            data_in = edge
            tin = [int(dtypes.string), int(dtypes.int64), dtype_enum]
        parts[src_worker].append(pbwire.node_def(
            pname, 'PyFunc', [kname, STEP_PH + ':0', data_in], '',
            {'token': ('s', PUBLISH_TOKEN),
             'Tin': ('list', {'type': tin}),
             'Tout': ('list', {'type': [int(dtypes.int32)]})}))
        edge_publishes[edge] = (src_worker, pname)
        return pname

    def _recv(dst_worker, src_worker, src_name, port, dtype_enum,
              is_control):
        edge = '%s:%d' % (src_name, port) if not is_control else \
            '^' + src_name
        k = (edge, dst_worker)
        if k in recv_added:
            return recv_added[k]
        addr = cluster[src_worker]
        key = '%s|%s' % (addr, edge)
        kname = _key_const(dst_worker, key)
        counters['n'] += 1
        rname = '__dist_recv_%d' % counters['n']
        out_t = dtype_enum if not is_control else int(dtypes.int32)
        parts[dst_worker].append(pbwire.node_def(
            rname, 'PyFunc', [kname, STEP_PH + ':0'], '',
            {'token': ('s', RECV_TOKEN),
             'Tin': ('list', {'type': [int(dtypes.string),
                                       int(dtypes.int64)]}),
             'Tout': ('list', {'type': [out_t]})}))
        recv_added[k] = rname
        return rname

    by_name = {nd['name']: nd for _, nd in nodes}
    publish_names = {w: [] for w in workers}
    for raw, nd in nodes:
        w = owner[nd['name']]
        new_inputs = []
        changed = False
        for inp in nd['input']:
            if inp.startswith('^'):
                src = inp[1:]
                sw = owner[src]
                if sw != w:
                    p = _publish(sw, src, 0, 0, True)
                    if p:
                        publish_names[sw].append(p)
                    r = _recv(w, sw, src, 0, 0, True)
                    new_inputs.append('^' + r)
                    changed = True
                    continue
                new_inputs.append(inp)
                continue
            name, _, port_s = inp.partition(':')
            port = int(port_s or 0)
            sw = owner[name]
            if sw != w:
                dt = out_dtypes[name][port]
                if out_refs.get(name, [False])[port]:
                    raise ValueError('ref edge %s crosses workers' % inp)
                p = _publish(sw, name, port, dt, False)
                if p:
                    publish_names[sw].append(p)
                r = _recv(w, sw, name, port, dt, False)
                new_inputs.append(r + ':0')
                changed = True
            else:
                new_inputs.append(inp)
        if changed:
            # re-emit the node with rewritten inputs and its local device
            attrs = {k: v for k, v in nd['attr'].items()}
            nb = pbwire.node_def(nd['name'], nd['op'], new_inputs,
                                 _local_device(nd['device']), attrs)
            parts[w].append(nb)
        else:
            parts[w].append(raw)

    # per-partition step-id placeholder
    for w in workers:
        parts[w].append(pbwire.node_def(
            STEP_PH, 'Placeholder', [], '',
            {'dtype': ('type', int(dtypes.int64)),
             'shape': ('shape', [])}))

    result = {w: pbwire.graph_def(parts[w]) for w in workers}
    meta = {
        'owner': owner,
        'publishes': {w: publish_names[w] for w in workers},
        'edge_publishes': dict(edge_publishes),
    }
    return result, meta
