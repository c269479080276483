"""Distributed runtime: ClusterSpec + tf.train.Server + grpc:// sessions.

Capability analog of the reference's grpc distributed runtime (SURVEY.md
§2.3: server_lib.py:223, GrpcServer, MasterService CreateSession/RunStep).
The gRPC transport here is python-grpcio with a bytes-level protocol (JSON
header + raw tensor payloads — the same zero-copy-tensor idea as the
reference's grpc_tensor_coding.cc, without a protoc dependency).

Round-1 scope note: a grpc:// session runs the full graph on the target
server (master==worker); cross-worker graph splits with remote rendezvous are
a round-2 item. Single-node multi-GPU scaling uses the first-class RCCL path
(parallel/dist.py) instead of PS sharding.
"""
import json
import struct
import threading
from concurrent import futures

import numpy as np

import grpc

_RUNSTEP = '/stf.MasterService/RunStep'
_CREATE = '/stf.MasterService/CreateSession'
_EXTEND = '/stf.MasterService/ExtendSession'
_REGISTER = '/stf.WorkerService/RegisterGraph'
_RUNGRAPH = '/stf.WorkerService/RunGraph'
_RECVTENSOR = '/stf.WorkerService/RecvTensor'
_CLEANUP = '/stf.WorkerService/CleanupStep'

# process-global tensor store for cross-worker edges: several in-process
# Servers (the localhost test cluster) share it, keyed by producer address.
_dist_store = {}
_dist_store_cv = threading.Condition()
_dist_channels = {}
_dist_channels_lock = threading.Lock()


def _dist_channel(addr):
    with _dist_channels_lock:
        ch = _dist_channels.get(addr)
        if ch is None:
            ch = grpc.insecure_channel(
                addr, options=[('grpc.max_receive_message_length', 1 << 30),
                               ('grpc.max_send_message_length', 1 << 30)])
            _dist_channels[addr] = ch
        return ch


def _dist_publish_fn(key, step, value):
    """PyFunc callable on the producer worker: publish a cross-worker edge
    tensor for this step (the send half of the reference's rendezvous)."""
    k = (bytes(key).decode(), int(step))
    with _dist_store_cv:
        _dist_store[k] = value
        _dist_store_cv.notify_all()
    return np.int32(0)


def _dist_recv_fn(key, step):
    """PyFunc callable on the consumer worker: pull the edge tensor from the
    producer via RecvTensor (rpc_rendezvous_mgr.cc:273 pull model)."""
    key = bytes(key).decode()
    addr = key.split('|', 1)[0]
    meta = json.dumps({'key': key, 'step': int(step)}).encode()
    stub = _dist_channel(addr).unary_unary(_RECVTENSOR)
    resp = stub(meta, timeout=120)
    out = _unpack_tensors(resp)
    return out[0][1]


class ClusterSpec(object):
    def __init__(self, cluster):
        if isinstance(cluster, ClusterSpec):
            self._cluster = dict(cluster._cluster)
        else:
            self._cluster = {k: list(v) for k, v in dict(cluster).items()}

    def as_dict(self):
        return dict(self._cluster)

    def jobs(self):
        return list(self._cluster)

    def job_tasks(self, job):
        return list(self._cluster[job])

    def task_address(self, job, task):
        return self._cluster[job][task]

    def num_tasks(self, job):
        return len(self._cluster[job])


def _pack_tensors(named_arrays):
    """[(name, np.ndarray|bytes)] -> framed bytes."""
    header = []
    payloads = []
    for name, arr in named_arrays:
        if isinstance(arr, (bytes, str)):
            b = arr.encode() if isinstance(arr, str) else arr
            header.append({'name': name, 'dtype': 'bytes', 'shape': [],
                           'nbytes': len(b)})
            payloads.append(b)
        else:
            arr = np.ascontiguousarray(arr)
            header.append({'name': name, 'dtype': arr.dtype.str,
                           'shape': list(arr.shape), 'nbytes': arr.nbytes})
            payloads.append(arr.tobytes())
    hj = json.dumps(header).encode()
    out = struct.pack('<I', len(hj)) + hj + b''.join(payloads)
    return out


def _unpack_tensors(data):
    (hlen,) = struct.unpack('<I', data[:4])
    header = json.loads(data[4:4 + hlen])
    off = 4 + hlen
    out = []
    for h in header:
        b = data[off:off + h['nbytes']]
        off += h['nbytes']
        if h['dtype'] == 'bytes':
            out.append((h['name'], b))
        else:
            out.append((h['name'],
                        np.frombuffer(b, dtype=np.dtype(h['dtype']))
                        .reshape(h['shape'])))
    return out


class Server(object):
    """tf.train.Server: in-process gRPC server hosting master+worker."""

    def __init__(self, server_or_cluster_def, job_name=None, task_index=0,
                 protocol='grpc', config=None, start=True):
        if isinstance(server_or_cluster_def, dict):
            cluster = ClusterSpec(server_or_cluster_def)
        elif isinstance(server_or_cluster_def, ClusterSpec):
            cluster = server_or_cluster_def
        else:
            cluster = ClusterSpec(server_or_cluster_def)
        self._cluster = cluster
        self._job = job_name or cluster.jobs()[0]
        self._task = task_index
        addr = cluster.task_address(self._job, task_index)
        self._addr = addr
        self._sessions = {}
        self._next_id = [0]
        self._lock = threading.Lock()

        def create_handler(req, ctx):
            from simple_tensorflow_amd import _core
            with self._lock:
                handle = 'sess_%d' % self._next_id[0]
                self._next_id[0] += 1
            self._sessions[handle] = _MasterSession(self, bytes(req))
            return handle.encode()

        def extend_handler(req, ctx):
            (hlen,) = struct.unpack('<I', req[:4])
            handle = req[4:4 + hlen].decode()
            self._sessions[handle].extend(bytes(req[4 + hlen:]))
            return b'ok'

        def run_handler(req, ctx):
            (hlen,) = struct.unpack('<I', req[:4])
            meta = json.loads(req[4:4 + hlen])
            feeds_raw = _unpack_tensors(req[4 + hlen:])
            feeds = {}
            for name, arr in feeds_raw:
                feeds[name] = arr if isinstance(arr, bytes) else arr
            entry = self._sessions[meta['handle']]
            if isinstance(entry, _MasterSession):
                results = entry.run(feeds, meta['fetches'], meta['targets'])
            else:
                results = entry.run(feeds, meta['fetches'], meta['targets'])
            return _pack_tensors([('r%d' % i, r)
                                  for i, r in enumerate(results)])

        # ---- worker service (GraphMgr analog, graph_mgr.cc:238) ----
        self._worker_graphs = {}

        def register_graph_handler(req, ctx):
            (hlen,) = struct.unpack('<I', req[:4])
            handle = req[4:4 + hlen].decode()
            from simple_tensorflow_amd import _core
            _ensure_dist_pyfuncs()
            if handle in self._worker_graphs:
                self._worker_graphs[handle].extend(bytes(req[4 + hlen:]))
            else:
                sess = _core.Session(False)
                sess.create(bytes(req[4 + hlen:]))
                self._worker_graphs[handle] = sess
            return b'ok'

        def run_graph_handler(req, ctx):
            (hlen,) = struct.unpack('<I', req[:4])
            meta = json.loads(req[4:4 + hlen])
            feeds_raw = _unpack_tensors(req[4 + hlen:])
            feeds = {name: arr for name, arr in feeds_raw}
            from simple_tensorflow_amd.python.training import graph_partition
            feeds[graph_partition.STEP_PH] = np.int64(meta['step'])
            sess = self._worker_graphs[meta['handle']]
            results = sess.run(feeds, meta['fetches'], meta['targets'])
            return _pack_tensors([('r%d' % i, r)
                                  for i, r in enumerate(results)])

        def recv_tensor_handler(req, ctx):
            meta = json.loads(bytes(req))
            k = (meta['key'], meta['step'])
            deadline = 115.0
            with _dist_store_cv:
                ok = _dist_store_cv.wait_for(lambda: k in _dist_store,
                                             timeout=deadline)
                if not ok:
                    ctx.abort(grpc.StatusCode.DEADLINE_EXCEEDED,
                              'tensor %r never produced' % (meta['key'],))
                val = _dist_store[k]
            return _pack_tensors([('t', val)])

        def cleanup_handler(req, ctx):
            meta = json.loads(bytes(req))
            step = meta['step']
            with _dist_store_cv:
                for k in [k for k in _dist_store if k[1] == step]:
                    del _dist_store[k]
            return b'ok'

        handlers = {
            _CREATE: grpc.unary_unary_rpc_method_handler(
                create_handler, request_deserializer=None,
                response_serializer=None),
            _EXTEND: grpc.unary_unary_rpc_method_handler(extend_handler),
            _RUNSTEP: grpc.unary_unary_rpc_method_handler(run_handler),
        }
        worker_handlers = {
            _REGISTER: grpc.unary_unary_rpc_method_handler(
                register_graph_handler),
            _RUNGRAPH: grpc.unary_unary_rpc_method_handler(
                run_graph_handler),
            _RECVTENSOR: grpc.unary_unary_rpc_method_handler(
                recv_tensor_handler),
            _CLEANUP: grpc.unary_unary_rpc_method_handler(cleanup_handler),
        }
        generic = grpc.method_handlers_generic_handler(
            'stf.MasterService',
            {k.split('/')[-1]: v for k, v in handlers.items()})
        generic_w = grpc.method_handlers_generic_handler(
            'stf.WorkerService',
            {k.split('/')[-1]: v for k, v in worker_handlers.items()})
        self._server = grpc.server(
            futures.ThreadPoolExecutor(max_workers=16),
            options=[('grpc.max_receive_message_length', 1 << 30),
                     ('grpc.max_send_message_length', 1 << 30)])
        self._server.add_generic_rpc_handlers((generic, generic_w))
        port = addr.split(':')[-1]
        self._server.add_insecure_port('0.0.0.0:' + port)
        if start:
            self.start()

    @property
    def target(self):
        return 'grpc://' + self._addr

    @property
    def server_def(self):
        return {'cluster': self._cluster.as_dict(), 'job_name': self._job,
                'task_index': self._task}

    def start(self):
        self._server.start()
        return self

    def join(self):
        self._server.wait_for_termination()

    def stop(self):
        self._server.stop(0)

    @staticmethod
    def create_local_server(start=True):
        import socket
        s = socket.socket()
        s.bind(('127.0.0.1', 0))
        port = s.getsockname()[1]
        s.close()
        return Server({'local': ['127.0.0.1:%d' % port]}, 'local', 0,
                      start=start)


class GrpcRemoteCore(object):
    """Drop-in for _core.Session speaking to a remote Server (the reference's
    GrpcSession, rpc/grpc_session.cc:39)."""

    def __init__(self, target):
        addr = target[len('grpc://'):]
        self._channel = grpc.insecure_channel(
            addr, options=[('grpc.max_receive_message_length', 1 << 30),
                           ('grpc.max_send_message_length', 1 << 30)])
        self._create = self._channel.unary_unary(_CREATE)
        self._extend = self._channel.unary_unary(_EXTEND)
        self._run = self._channel.unary_unary(_RUNSTEP)
        self._handle = None

    def create(self, graph_def_bytes):
        self._handle = self._create(graph_def_bytes).decode()

    def extend(self, graph_def_bytes):
        h = self._handle.encode()
        self._extend(struct.pack('<I', len(h)) + h + graph_def_bytes)

    def run(self, feeds, fetches, targets):
        meta = json.dumps({'handle': self._handle, 'fetches': list(fetches),
                           'targets': list(targets)}).encode()
        body = _pack_tensors(list(feeds.items()))
        resp = self._run(struct.pack('<I', len(meta)) + meta + body)
        out = [arr for _, arr in _unpack_tensors(resp)]
        return out

    def sync(self):
        pass

    def num_gpus(self):
        return 0


def _ensure_dist_pyfuncs():
    from simple_tensorflow_amd import _core
    _core.register_py_func('__dist_publish', _dist_publish_fn)
    _core.register_py_func('__dist_recv', _dist_recv_fn)


class _MasterSession(object):
    """Master-side session: prunes the client graph to each run signature,
    partitions it by worker (/job:X/task:N device specs), registers each
    partition with the owning worker (RegisterGraph) and fans RunGraph out
    in parallel per step — the reference's MasterSession::Run /
    BuildAndRegisterPartitions / RunPartitions (master_session.cc:1173,
    1199,512) re-shaped onto this framework's python-grpcio transport.
    Single-worker graphs fall through to a plain local session."""

    def __init__(self, server, graph_bytes):
        self._server = server
        self._graph = graph_bytes
        self._step = [0]
        self._lock = threading.Lock()
        self._dirty = True
        self._handles = {}
        self._registered = {}
        self._local = None
        self._local_len = 0
        cluster = server._cluster
        self._addrs = {}
        for job in cluster.jobs():
            for t, addr in enumerate(cluster.job_tasks(job)):
                self._addrs[(job, t)] = addr
        self._default = (server._job, server._task)
        self.is_distributed = True  # decided per run; master always wraps

    def _spans_workers(self, graph_bytes):
        from simple_tensorflow_amd.python.training import graph_partition
        workers = set()
        for _, nd in graph_partition._parse_nodes_raw(graph_bytes):
            workers.add(graph_partition._worker_of(nd['device'],
                                                   self._default))
        return len(workers) > 1

    def extend(self, graph_bytes):
        with self._lock:
            self._graph += graph_bytes  # GraphDef concat == node concat
            self._dirty = True

    # ---- single-worker fall-through ----
    def _run_local(self, feeds, fetches, targets):
        from simple_tensorflow_amd import _core
        if self._local is None:
            self._local = _core.Session(False)
            self._local.create(self._graph)
            self._local_len = len(self._graph)
        elif self._local_len < len(self._graph):
            self._local.extend(self._graph[self._local_len:])
            self._local_len = len(self._graph)
        return self._local.run(feeds, fetches, targets)

    def _ensure_registered(self):
        """Partition the full accumulated graph and (re)register only the
        per-worker delta nodes — worker sessions persist variable state
        across signatures and extends (reference GraphMgr semantics)."""
        from simple_tensorflow_amd.python.training import graph_partition
        if not self._dirty:
            return
        parts, meta = graph_partition.partition_by_worker(
            self._graph, self._addrs, self._default)
        self._owner = meta['owner']
        self._edge_pubs = meta['edge_publishes']
        for w, gd in parts.items():
            nodes = graph_partition._parse_nodes_raw(gd)
            sent = self._registered.setdefault(w, set())
            delta = [raw for raw, nd in nodes if nd['name'] not in sent]
            if not delta:
                continue
            for raw, nd in nodes:
                sent.add(nd['name'])
            h = self._handles.get(w)
            if h is None:
                h = 'wg_%d_%s_%d' % (id(self), w[0], w[1])
                self._handles[w] = h
            stub = _dist_channel(self._addrs[w]).unary_unary(_REGISTER)
            hb = h.encode()
            stub(struct.pack('<I', len(hb)) + hb + pbwire_graph_def(delta),
                 timeout=120)
        self._nodes_cache = {
            nd['name']: nd
            for _, nd in graph_partition._parse_nodes_raw(self._graph)}
        self._dirty = False

    def _needed_publishes(self, roots):
        """Backward closure from the step's roots over the CLIENT graph;
        every cross-worker edge inside the closure needs its publish node
        targeted on the producer."""
        by_name = self._nodes_cache
        keep = set()
        stack = [r for r in roots if r in by_name]
        while stack:
            n = stack.pop()
            if n in keep:
                continue
            keep.add(n)
            for inp in by_name[n]['input']:
                stack.append(inp.lstrip('^').split(':')[0])
        pubs = {}
        for n in keep:
            w = self._owner[n]
            for inp in by_name[n]['input']:
                if inp.startswith('^'):
                    src = inp[1:]
                    if self._owner[src] != w:
                        key = '^' + src
                        pw, pn = self._edge_pubs[key]
                        pubs.setdefault(pw, set()).add(pn)
                    continue
                name, _, port_s = inp.partition(':')
                if self._owner[name] != w:
                    key = '%s:%d' % (name, int(port_s or 0))
                    pw, pn = self._edge_pubs[key]
                    pubs.setdefault(pw, set()).add(pn)
        return pubs

    def run(self, feeds, fetches, targets):
        def node_of(name):
            return name.split(':')[0].lstrip('^')

        with self._lock:
            if not self._spans_workers(self._graph):
                return self._run_local(feeds, fetches, targets)
            self._ensure_registered()
            roots = [node_of(n) for n in list(fetches) + list(targets)]
            pubs = self._needed_publishes(roots)
            handles = dict(self._handles)
            owner = self._owner
            self._step[0] += 1
            step = self._step[0]

        per = {w: {'feeds': {}, 'fetches': [], 'targets': []}
               for w in handles}
        for name, val in feeds.items():
            per[owner[node_of(name)]]['feeds'][name] = val
        fetch_slots = []
        for name in fetches:
            w = owner[node_of(name)]
            fetch_slots.append((w, len(per[w]['fetches'])))
            per[w]['fetches'].append(name)
        for name in targets:
            per[owner[node_of(name)]]['targets'].append(name)
        for w, pnames in pubs.items():
            per[w]['targets'] = list(per[w]['targets']) + sorted(pnames)
        active = [w for w in handles
                  if per[w]['feeds'] or per[w]['fetches'] or
                  per[w]['targets']]

        results = {}
        errors_ = []

        def run_one(w):
            try:
                meta = json.dumps({
                    'handle': handles[w], 'step': step,
                    'fetches': per[w]['fetches'],
                    'targets': per[w]['targets']}).encode()
                body = _pack_tensors(list(per[w]['feeds'].items()))
                stub = _dist_channel(self._addrs[w]).unary_unary(_RUNGRAPH)
                resp = stub(struct.pack('<I', len(meta)) + meta + body,
                            timeout=180)
                results[w] = [a for _, a in _unpack_tensors(resp)]
            except Exception as e:  # noqa: BLE001 - collected and re-raised
                errors_.append(e)

        threads = [threading.Thread(target=run_one, args=(w,))
                   for w in active]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        # release this step's published tensors on every worker
        for w in active:
            try:
                _dist_channel(self._addrs[w]).unary_unary(_CLEANUP)(
                    json.dumps({'step': step}).encode(), timeout=30)
            except Exception:  # noqa: BLE001 - cleanup is best-effort
                pass
        if errors_:
            raise errors_[0]
        return [results[w][i] for (w, i) in fetch_slots]


def pbwire_graph_def(node_bytes_list):
    from simple_tensorflow_amd.python.framework import pbwire
    return pbwire.graph_def(node_bytes_list)
