"""Metrics instrumentation: Counter / IntGauge / Sampler cells collected
through a process-global CollectionRegistry.

Capability analog of the reference's core/lib/monitoring
(counter.h, gauge.h, sampler.h, collection_registry.h:124): metric objects
are registered at construction under a unique '/path/name', hold per-label
cells, and a collector snapshots every registered metric. Python-native
redesign — the hot path here is framework bookkeeping, not kernels, so there
is no C++ mirror; thread-safety via a single registry lock.
"""
import bisect
import threading


class _Cell(object):
    __slots__ = ('_value', '_lock')

    def __init__(self, value):
        self._value = value
        self._lock = threading.Lock()

    def value(self):
        with self._lock:
            return self._value


class CounterCell(_Cell):
    def increment_by(self, step):
        if step < 0:
            raise ValueError('Counter can only increase')
        with self._lock:
            self._value += step

    def increment(self):
        self.increment_by(1)


class GaugeCell(_Cell):
    def set(self, value):
        with self._lock:
            self._value = value


class SamplerCell(object):
    """Histogram cell over explicit bucket boundaries."""

    def __init__(self, buckets):
        self._buckets = list(buckets)
        self._counts = [0] * (len(self._buckets) + 1)
        self._sum = 0.0
        self._num = 0
        self._lock = threading.Lock()

    def add(self, sample):
        with self._lock:
            self._counts[bisect.bisect_right(self._buckets, sample)] += 1
            self._sum += sample
            self._num += 1

    def value(self):
        with self._lock:
            return {'buckets': list(self._buckets),
                    'counts': list(self._counts),
                    'sum': self._sum, 'num': self._num}


class _Metric(object):
    def __init__(self, name, description, label_names, make_cell):
        if len(label_names) > 2:
            raise ValueError('at most 2 labels supported')
        self.name = name
        self.description = description
        self._label_names = tuple(label_names)
        self._make_cell = make_cell
        self._cells = {}
        self._lock = threading.Lock()
        CollectionRegistry.default().register(self)

    def get_cell(self, *labels):
        if len(labels) != len(self._label_names):
            raise ValueError('expected %d labels, got %d'
                             % (len(self._label_names), len(labels)))
        with self._lock:
            cell = self._cells.get(labels)
            if cell is None:
                cell = self._cells[labels] = self._make_cell()
            return cell

    def snapshot(self):
        with self._lock:
            return {labels: cell.value() for labels, cell in
                    self._cells.items()}


class Counter(_Metric):
    """Monotonic int64 counter, e.g. Counter('/stf/session/runs', '...')."""

    def __init__(self, name, description='', *label_names):
        super().__init__(name, description, label_names,
                         lambda: CounterCell(0))


class IntGauge(_Metric):
    def __init__(self, name, description='', *label_names):
        super().__init__(name, description, label_names,
                         lambda: GaugeCell(0))


class StringGauge(_Metric):
    def __init__(self, name, description='', *label_names):
        super().__init__(name, description, label_names,
                         lambda: GaugeCell(''))


class Sampler(_Metric):
    def __init__(self, name, buckets, description='', *label_names):
        super().__init__(name, description, label_names,
                         lambda: SamplerCell(buckets))


def exponential_buckets(scale, growth, count):
    return [scale * growth ** i for i in range(count)]


class CollectionRegistry(object):
    _instance = None
    _instance_lock = threading.Lock()

    def __init__(self):
        self._metrics = {}
        self._lock = threading.Lock()

    @classmethod
    def default(cls):
        with cls._instance_lock:
            if cls._instance is None:
                cls._instance = cls()
            return cls._instance

    def register(self, metric):
        with self._lock:
            if metric.name in self._metrics:
                raise ValueError('metric %r already registered' % metric.name)
            self._metrics[metric.name] = metric

    def unregister(self, metric):
        with self._lock:
            self._metrics.pop(metric.name, None)

    def collect_metrics(self):
        """Snapshot of every registered metric: {name: {labels: value}}."""
        with self._lock:
            metrics = list(self._metrics.values())
        return {m.name: m.snapshot() for m in metrics}
