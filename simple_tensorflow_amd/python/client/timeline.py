"""RunOptions / RunMetadata / StepStats / Timeline — the tracing surface
(analog of reference python/client/timeline.py:346 + config.proto RunOptions
and step_stats.proto). StepStats serializes byte-compatibly with the
reference's step_stats.proto (DeviceStepStats device=1 node_stats=2;
NodeExecStats node_name=1 all_start_micros=2 op_start_rel=3 op_end_rel=4
all_end_rel=5 timeline_label=8)."""
import json

from simple_tensorflow_amd.python.framework.pbwire import f_bytes, f_varint


class RunOptions(object):
    NO_TRACE = 0
    SOFTWARE_TRACE = 1
    HARDWARE_TRACE = 2
    FULL_TRACE = 3

    def __init__(self, trace_level=0, timeout_in_ms=0,
                 inter_op_thread_pool=0, output_partition_graphs=False):
        self.trace_level = trace_level
        self.timeout_in_ms = timeout_in_ms
        self.inter_op_thread_pool = inter_op_thread_pool
        self.output_partition_graphs = output_partition_graphs


class RunMetadata(object):
    def __init__(self):
        self.step_stats = None
        self.partition_graphs = []


class NodeExecStats(object):
    __slots__ = ('node_name', 'op', 'all_start_micros', 'op_end_rel_micros')

    def __init__(self, node_name, op, all_start_micros, op_end_rel_micros):
        self.node_name = node_name
        self.op = op
        self.all_start_micros = all_start_micros
        self.op_end_rel_micros = op_end_rel_micros


class StepStats(object):
    """Wraps the pybind (node, op, start_us, end_us[, device]) tuples.

    Entries with an empty device are the host enqueue lane; entries tagged
    by the GpuTracer (csrc/gpu/gpu_tracer.cc) land in their own
    /device:GPU:n/stream:compute lane with hardware kernel intervals."""

    def __init__(self, raw):
        lanes = {}
        for entry in raw:
            dev = entry[4] if len(entry) > 4 and entry[4] else \
                '/job:localhost/replica:0/task:0'
            lanes.setdefault(dev, []).append(entry[:4])
        if not lanes:
            lanes['/job:localhost/replica:0/task:0'] = []
        self.dev_stats = [DeviceStepStats(dev, entries)
                          for dev, entries in sorted(lanes.items())]

    def SerializeToString(self):
        out = b''
        for ds in self.dev_stats:
            body = f_bytes(1, ds.device)
            for ns in ds.node_stats:
                nb = f_bytes(1, ns.node_name)
                nb += f_varint(2, ns.all_start_micros)
                nb += f_varint(3, 0)
                nb += f_varint(4, ns.op_end_rel_micros)
                nb += f_varint(5, ns.op_end_rel_micros)
                nb += f_bytes(8, ns.op)
                body += f_bytes(2, nb)
            out += f_bytes(1, body)
        return out


class DeviceStepStats(object):
    def __init__(self, device, raw):
        self.device = device
        self.node_stats = [
            NodeExecStats(node, op, start, max(0, end - start))
            for (node, op, start, end) in raw]


class Timeline(object):
    """Chrome-trace-format generation from StepStats (reference
    python/client/timeline.py Timeline.generate_chrome_trace_format)."""

    def __init__(self, step_stats, graph=None):
        self._step_stats = step_stats

    def generate_chrome_trace_format(self, show_dataflow=True,
                                     show_memory=False):
        events = []
        pid = 0
        for ds in self._step_stats.dev_stats:
            events.append({'name': 'process_name', 'ph': 'M', 'pid': pid,
                           'args': {'name': ds.device}})
            for ns in ds.node_stats:
                events.append({
                    'name': ns.node_name, 'cat': 'Op', 'ph': 'X',
                    'ts': ns.all_start_micros,
                    'dur': max(1, ns.op_end_rel_micros),
                    'pid': pid, 'tid': 0,
                    'args': {'op': ns.op}})
            pid += 1
        return json.dumps({'traceEvents': events})
