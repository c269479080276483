"""simple_tensorflow_amd — an MI355X-native dataflow DL framework with the
capability surface of TF 1.0.

Use `import simple_tensorflow_amd as tf` — the module exposes the familiar
tf.* names (Session, Graph, Variable, train.*, nn.*, ...).
"""
import os as _os
import sys as _sys

# The pybind extension must be importable as simple_tensorflow_amd._core.
from simple_tensorflow_amd import _core  # noqa: F401

from simple_tensorflow_amd.python.framework import dtypes as _dtypes
from simple_tensorflow_amd.python.framework import ops as _ops
from simple_tensorflow_amd.python.client import session as _session
from simple_tensorflow_amd.python.ops import (  # noqa: F401
    array_ops as _array_ops,
    linalg_ops as _linalg_ops,
    spectral_ops as _spectral_ops,
    string_ops as _string_ops,
    functional_ops as _functional_ops,
    clip_ops as _clip_ops,
    control_flow_ops as _control_flow_ops,
    gradients_impl as _gradients_impl,
    init_ops as _init_ops,
    math_ops as _math_ops,
    nn_ops as _nn,
    random_ops as _random_ops,
    state_ops as _state_ops,
    variables as _variables,
)
from simple_tensorflow_amd.python.training import optimizer as _optimizer
from simple_tensorflow_amd.python.training import sync_replicas_optimizer as _sync_opt
from simple_tensorflow_amd.python.training import training_util as _training_util
from simple_tensorflow_amd.python.training import saver as _saver
from simple_tensorflow_amd.python.training import coordinator as _coord
from simple_tensorflow_amd.python.training import input as _input
from simple_tensorflow_amd.python.training import monitored_session as _ms
from simple_tensorflow_amd.python.training import server_lib as _server_lib
from simple_tensorflow_amd.python.training import device_setter as _device_setter
from simple_tensorflow_amd.python.summary import summary as _summary_mod
from simple_tensorflow_amd.python.summary import writer as _summary_writer
from simple_tensorflow_amd.python.framework import errors  # noqa: F401

# ---- dtypes ----
float32 = _dtypes.float32
float64 = _dtypes.float64
double = _dtypes.float64
int32 = _dtypes.int32
int64 = _dtypes.int64
int16 = _dtypes.int16
int8 = _dtypes.int8
uint8 = _dtypes.uint8
uint16 = _dtypes.uint16
bool = _dtypes.bool  # noqa: A001
string = _dtypes.string
bfloat16 = _dtypes.bfloat16
float16 = _dtypes.float16
half = _dtypes.float16
DType = _dtypes.DType
as_dtype = _dtypes.as_dtype

# ---- framework ----
Graph = _ops.Graph
Operation = _ops.Operation
Tensor = _ops.Tensor
TensorShape = _ops.TensorShape
GraphKeys = _ops.GraphKeys
get_default_graph = _ops.get_default_graph
reset_default_graph = _ops.reset_default_graph
device = _ops.device
name_scope = _ops.name_scope
control_dependencies = _ops.control_dependencies
colocate_with = _ops.colocate_with
convert_to_tensor = _ops.convert_to_tensor
constant = _ops.constant
from simple_tensorflow_amd.python.framework import importer as _importer
import_graph_def = _importer.import_graph_def
RegisterGradient = _ops.RegisterGradient
NoGradient = _ops.NoGradient
NotDifferentiable = _ops.NotDifferentiable
add_to_collection = lambda name, value: get_default_graph().add_to_collection(name, value)  # noqa: E731
get_collection = lambda name, scope=None: get_default_graph().get_collection(name, scope)  # noqa: E731
get_collection_ref = lambda name: get_default_graph().get_collection_ref(name)  # noqa: E731

# ---- session ----
Session = _session.Session
InteractiveSession = _session.InteractiveSession
get_default_session = _session.get_default_session

# ---- variables ----
Variable = _variables.Variable
global_variables = _variables.global_variables
all_variables = _variables.all_variables
trainable_variables = _variables.trainable_variables
local_variables = _variables.local_variables
moving_average_variables = _variables.moving_average_variables
global_variables_initializer = _variables.global_variables_initializer
initialize_all_variables = _variables.initialize_all_variables
local_variables_initializer = _variables.local_variables_initializer
variables_initializer = _variables.variables_initializer
is_variable_initialized = _variables.is_variable_initialized
assign = _state_ops.assign
assign_add = _state_ops.assign_add
assign_sub = _state_ops.assign_sub
scatter_add = _state_ops.scatter_add
scatter_sub = _state_ops.scatter_sub

# ---- array ops ----
placeholder = _array_ops.placeholder
identity = _array_ops.identity
stop_gradient = _array_ops.stop_gradient
shape = _array_ops.shape
shape_n = _array_ops.shape_n
rank = _array_ops.rank
size = _array_ops.size
reshape = _array_ops.reshape
expand_dims = _array_ops.expand_dims
squeeze = _array_ops.squeeze
zeros = _array_ops.zeros
ones = _array_ops.ones
fill = _array_ops.fill
zeros_like = _array_ops.zeros_like
ones_like = _array_ops.ones_like
concat = _array_ops.concat
split = _array_ops.split
stack = _array_ops.stack
pack = _array_ops.pack
unstack = _array_ops.unstack
unpack = _array_ops.unpack
slice = _array_ops.slice  # noqa: A001
pad = _array_ops.pad
transpose = _array_ops.transpose
gather = _array_ops.gather
tile = _array_ops.tile
one_hot = _array_ops.one_hot
where = _array_ops.where
check_numerics = _array_ops.check_numerics
unsorted_segment_sum = _array_ops.unsorted_segment_sum

# ---- math ----
add = _math_ops.add
subtract = _math_ops.subtract
sub = _math_ops.sub
multiply = _math_ops.multiply
mul = _math_ops.mul
divide = _math_ops.divide
div = _math_ops.div
truediv = _math_ops.truediv
floordiv = _math_ops.floordiv
mod = _math_ops.mod
pow = _math_ops.pow  # noqa: A001
maximum = _math_ops.maximum
minimum = _math_ops.minimum
squared_difference = _math_ops.squared_difference
less = _math_ops.less
less_equal = _math_ops.less_equal
greater = _math_ops.greater
greater_equal = _math_ops.greater_equal
equal = _math_ops.equal
not_equal = _math_ops.not_equal
logical_and = _math_ops.logical_and
logical_or = _math_ops.logical_or
logical_not = _math_ops.logical_not
negative = _math_ops.negative
neg = _math_ops.neg
abs = _math_ops.abs  # noqa: A001
sign = _math_ops.sign
square = _math_ops.square
sqrt = _math_ops.sqrt
rsqrt = _math_ops.rsqrt
exp = _math_ops.exp
log = _math_ops.log
log1p = _math_ops.log1p
tanh = _math_ops.tanh
sigmoid = _math_ops.sigmoid
sin = _math_ops.sin
cos = _math_ops.cos
floor = _math_ops.floor
ceil = _math_ops.ceil
round = _math_ops.round  # noqa: A001
reciprocal = _math_ops.reciprocal
is_nan = _math_ops.is_nan
is_inf = _math_ops.is_inf
is_finite = _math_ops.is_finite
cast = _math_ops.cast
to_float = _math_ops.to_float
to_double = _math_ops.to_double
to_int32 = _math_ops.to_int32
to_int64 = _math_ops.to_int64
matmul = _math_ops.matmul
batch_matmul = _math_ops.batch_matmul
add_n = _math_ops.add_n
reduce_sum = _math_ops.reduce_sum
reduce_mean = _math_ops.reduce_mean
reduce_max = _math_ops.reduce_max
reduce_min = _math_ops.reduce_min
reduce_prod = _math_ops.reduce_prod
reduce_all = _math_ops.reduce_all
reduce_any = _math_ops.reduce_any
argmax = _math_ops.argmax
argmin = _math_ops.argmin
select = _math_ops.select
range = _math_ops.range  # noqa: A001
cumsum = _math_ops.cumsum
realdiv = _math_ops.realdiv
div_no_nan = _math_ops.div_no_nan
norm = _math_ops.norm
tensordot = _math_ops.tensordot
trace = _math_ops.trace
placeholder_with_default = _array_ops.placeholder_with_default
eye = _array_ops.eye
meshgrid = _array_ops.meshgrid
reverse = _array_ops.reverse
reverse_v2 = _array_ops.reverse_v2
unique = _array_ops.unique
unique_with_counts = _array_ops.unique_with_counts
setdiff1d = _array_ops.setdiff1d
dynamic_partition = _array_ops.dynamic_partition
gather_nd = _array_ops.gather_nd
scatter_nd = _array_ops.scatter_nd
diag = _array_ops.diag
diag_part = _array_ops.diag_part
matrix_diag = _array_ops.matrix_diag
matrix_diag_part = _array_ops.matrix_diag_part
matrix_set_diag = _array_ops.matrix_set_diag
matrix_band_part = _array_ops.matrix_band_part
space_to_depth = _array_ops.space_to_depth
depth_to_space = _array_ops.depth_to_space
mirror_pad = _array_ops.mirror_pad
reverse_sequence = _array_ops.reverse_sequence
bitcast = _array_ops.bitcast

# linear algebra (reference tf.linalg / root exports)
cholesky = _linalg_ops.cholesky
matrix_determinant = _linalg_ops.matrix_determinant
matrix_inverse = _linalg_ops.matrix_inverse
matrix_solve = _linalg_ops.matrix_solve
matrix_triangular_solve = _linalg_ops.matrix_triangular_solve
matrix_solve_ls = _linalg_ops.matrix_solve_ls
qr = _linalg_ops.qr
svd = _linalg_ops.svd
self_adjoint_eig = _linalg_ops.self_adjoint_eig
self_adjoint_eigvals = _linalg_ops.self_adjoint_eigvals
eye = _linalg_ops.eye

# spectral (reference tf.fft / tf.spectral)
fft = _spectral_ops.fft
ifft = _spectral_ops.ifft
fft2d = _spectral_ops.fft2d
ifft2d = _spectral_ops.ifft2d
fft3d = _spectral_ops.fft3d
ifft3d = _spectral_ops.ifft3d
complex = _spectral_ops.complex
real = _spectral_ops.real
imag = _spectral_ops.imag
conj = _spectral_ops.conj
spectral = _spectral_ops

# strings (reference tf.string_* root exports)
string_join = _string_ops.string_join
string_split = _string_ops.string_split
substr = _string_ops.substr
string_to_hash_bucket = _string_ops.string_to_hash_bucket
string_to_hash_bucket_fast = _string_ops.string_to_hash_bucket_fast
string_to_hash_bucket_strong = _string_ops.string_to_hash_bucket_strong
string_to_number = _string_ops.string_to_number
reduce_join = _string_ops.reduce_join
encode_base64 = _string_ops.encode_base64
decode_base64 = _string_ops.decode_base64
strings = _string_ops

# functional ops (reference tf.map_fn / tf.foldl / tf.foldr / tf.scan)
map_fn = _functional_ops.map_fn
foldl = _functional_ops.foldl
foldr = _functional_ops.foldr
scan = _functional_ops.scan

from simple_tensorflow_amd.python.ops import special_math_ops as _sm  # noqa: E402
einsum = _sm.einsum

from simple_tensorflow_amd.python.ops import more_ops as _more  # noqa: E402
accumulate_n = _more.accumulate_n
boolean_mask = _more.boolean_mask
sequence_mask = _more.sequence_mask
multinomial = _more.multinomial
sparse_matmul = _more.sparse_matmul
space_to_batch = _more.space_to_batch
batch_to_space = _more.batch_to_space
required_space_to_batch_paddings = _more.required_space_to_batch_paddings
from simple_tensorflow_amd.python.ops import check_ops as _check_ops  # noqa: E402
Assert = _check_ops.Assert
Print = _check_ops.Print
assert_equal = _check_ops.assert_equal
assert_less = _check_ops.assert_less
assert_greater = _check_ops.assert_greater
assert_positive = _check_ops.assert_positive
assert_non_negative = _check_ops.assert_non_negative
add_check_numerics_ops = _check_ops.add_check_numerics_ops
confusion_matrix = _more.confusion_matrix
bincount = _more.bincount
random_shuffle = _more.random_shuffle
random_gamma = _more.random_gamma
tables_initializer = _more.tables_initializer
sparse_placeholder = _more.sparse_placeholder
from simple_tensorflow_amd.python.ops import parsing_ops as _parsing  # noqa: E402
parse_example = _parsing.parse_example
parse_single_example = _parsing.parse_single_example
FixedLenFeature = _parsing.FixedLenFeature
VarLenFeature = _parsing.VarLenFeature
floormod = _math_ops.mod
floor_div = _math_ops.floordiv if hasattr(_math_ops, 'floordiv') else None
truncatemod = _math_ops.truncatemod if hasattr(_math_ops, 'truncatemod') \
    else _math_ops.mod
_nn.accumulate_n = _more.accumulate_n
_nn.crelu = _more.crelu
_nn.zero_fraction = _more.zero_fraction
_nn.weighted_cross_entropy_with_logits = \
    _more.weighted_cross_entropy_with_logits
_nn.conv1d = _more.conv1d
_nn.separable_conv2d = _more.separable_conv2d
_nn.atrous_conv2d = _more.atrous_conv2d
_nn.bincount = _more.bincount
_nn.sufficient_statistics = _more.sufficient_statistics
_nn.normalize_moments = _more.normalize_moments
cumprod = _math_ops.cumprod
tan = _math_ops.tan
asin = _math_ops.asin
acos = _math_ops.acos
atan = _math_ops.atan
erf = _math_ops.erf
erfc = _math_ops.erfc
expm1 = _math_ops.expm1
lgamma = _math_ops.lgamma
digamma = _math_ops.digamma
rint = _math_ops.rint
mod = _math_ops.mod
approximate_equal = _math_ops.approximate_equal
as_string = lambda x, name=None: _ops.apply_op('AsString', _ops.convert_to_tensor(x), name=name)  # noqa: E731
decode_raw = lambda bytes_, out_type, name=None: _ops.apply_op('DecodeRaw', _ops.convert_to_tensor(bytes_), out_type=_dtypes.as_dtype(out_type), name=name)  # noqa: E731
segment_sum = _math_ops.segment_sum
segment_mean = _math_ops.segment_mean
segment_max = _math_ops.segment_max
segment_min = _math_ops.segment_min
segment_prod = _math_ops.segment_prod
case = _control_flow_ops.case
global_norm = _clip_ops.global_norm
clip_by_value = _clip_ops.clip_by_value
clip_by_norm = _clip_ops.clip_by_norm
clip_by_global_norm = _clip_ops.clip_by_global_norm

# ---- random ----
random_uniform = _random_ops.random_uniform
random_normal = _random_ops.random_normal
truncated_normal = _random_ops.truncated_normal
set_random_seed = _random_ops.set_random_seed

# ---- control flow ----
group = _control_flow_ops.group
no_op = _control_flow_ops.no_op
cond = _control_flow_ops.cond
while_loop = _control_flow_ops.while_loop
Assert = _control_flow_ops.Assert

# ---- gradients ----
gradients = _gradients_impl.gradients

# ---- initializers ----
zeros_initializer = _init_ops.zeros_initializer
ones_initializer = _init_ops.ones_initializer
constant_initializer = _init_ops.constant_initializer
random_uniform_initializer = _init_ops.random_uniform_initializer
random_normal_initializer = _init_ops.random_normal_initializer
truncated_normal_initializer = _init_ops.truncated_normal_initializer


# ---- nn / train sub-namespaces ----
class _NnModule(object):
    pass


nn = _nn  # module with conv2d/relu/softmax/...
from simple_tensorflow_amd.python.ops import rnn_cell_impl as _rnn  # noqa: E402
from simple_tensorflow_amd.python.ops import tensor_array_ops as _ta_ops  # noqa: E402
TensorArray = _ta_ops.TensorArray
from simple_tensorflow_amd.python.ops import variable_scope as _vs  # noqa: E402
variable_scope = _vs.variable_scope
get_variable = _vs.get_variable
get_variable_scope = _vs.get_variable_scope
VariableScope = _vs.VariableScope
nn.rnn_cell = _rnn
nn.dynamic_rnn = _rnn.dynamic_rnn
nn.static_rnn = _rnn.static_rnn
from simple_tensorflow_amd.python.ops import candidate_sampling_ops as _cs  # noqa: E402
nn.sampled_softmax_loss = _cs.sampled_softmax_loss
nn.nce_loss = _cs.nce_loss
nn.uniform_candidate_sampler = _cs.uniform_candidate_sampler
nn.log_uniform_candidate_sampler = _cs.log_uniform_candidate_sampler
nn.learned_unigram_candidate_sampler = _cs.learned_unigram_candidate_sampler
nn.compute_accidental_hits = _cs.compute_accidental_hits
nn.rnn = _rnn.static_rnn


class _TrainModule(object):
    Optimizer = _optimizer.Optimizer
    GradientDescentOptimizer = _optimizer.GradientDescentOptimizer
    MomentumOptimizer = _optimizer.MomentumOptimizer
    AdamOptimizer = _optimizer.AdamOptimizer
    RMSPropOptimizer = _optimizer.RMSPropOptimizer
    AdagradOptimizer = _optimizer.AdagradOptimizer
    AdadeltaOptimizer = _optimizer.AdadeltaOptimizer
    FtrlOptimizer = _optimizer.FtrlOptimizer
    ProximalGradientDescentOptimizer = \
        _optimizer.ProximalGradientDescentOptimizer
    create_global_step = staticmethod(_training_util.create_global_step)
    get_global_step = staticmethod(_training_util.get_global_step)
    get_or_create_global_step = staticmethod(
        _training_util.get_or_create_global_step)
    exponential_decay = staticmethod(_training_util.exponential_decay)
    polynomial_decay = staticmethod(_training_util.polynomial_decay)
    piecewise_constant = staticmethod(_training_util.piecewise_constant)
    natural_exp_decay = staticmethod(_training_util.natural_exp_decay)
    inverse_time_decay = staticmethod(_training_util.inverse_time_decay)

    @staticmethod
    def global_step(sess, global_step_tensor):
        return int(sess.run(global_step_tensor))

    @staticmethod
    def write_graph(graph_or_graph_def, logdir, name, as_text=True):
        import os
        os.makedirs(logdir, exist_ok=True)
        gd = graph_or_graph_def.as_graph_def() \
            if hasattr(graph_or_graph_def, 'as_graph_def') \
            else graph_or_graph_def
        path = os.path.join(logdir, name)
        with open(path, 'wb') as f:
            f.write(gd if isinstance(gd, (bytes, bytearray)) else bytes(gd))
        return path

    @staticmethod
    def summary_iterator(path):
        from simple_tensorflow_amd.python.lib.io import tf_record
        from simple_tensorflow_amd.python.summary import summary as _s
        for rec in tf_record.tf_record_iterator(path):
            yield _s.parse_event(rec) if hasattr(_s, 'parse_event') else rec

    @staticmethod
    def limit_epochs(tensor, num_epochs=None, name=None):
        from simple_tensorflow_amd.python.training import input as _inp
        return _inp.limit_epochs(tensor, num_epochs=num_epochs, name=name)

    @staticmethod
    def slice_input_producer(tensor_list, num_epochs=None, shuffle=True,
                             seed=None, capacity=32, name=None):
        from simple_tensorflow_amd.python.training import input as _inp
        return _inp.slice_input_producer(tensor_list,
                                         num_epochs=num_epochs,
                                         shuffle=shuffle, seed=seed,
                                         capacity=capacity, name=name)
    ExponentialMovingAverage = _training_util.ExponentialMovingAverage

    SyncReplicasOptimizer = _sync_opt.SyncReplicasOptimizer

    from simple_tensorflow_amd.python.lib import example_pb as _ex
    Example = _ex.Example
    Features = _ex.Features
    Feature = _ex.Feature
    BytesList = _ex.BytesList
    FloatList = _ex.FloatList
    Int64List = _ex.Int64List

    @staticmethod
    def Supervisor(*a, **kw):
        from simple_tensorflow_amd.python.training import supervisor
        return supervisor.Supervisor(*a, **kw)


    Saver = _saver.Saver

    @staticmethod
    def export_meta_graph(filename=None, **kw):
        from simple_tensorflow_amd.python.framework import meta_graph
        return meta_graph.export_meta_graph(filename=filename, **kw)

    @staticmethod
    def import_meta_graph(meta_graph_or_file, **kw):
        from simple_tensorflow_amd.python.framework import meta_graph
        return meta_graph.import_meta_graph(meta_graph_or_file, **kw)

    latest_checkpoint = staticmethod(_saver.latest_checkpoint)

    @staticmethod
    def NewCheckpointReader(prefix):
        from simple_tensorflow_amd.python.tools import inspect_checkpoint
        return inspect_checkpoint.NewCheckpointReader(prefix)

    get_checkpoint_state = staticmethod(_saver.get_checkpoint_state)
    update_checkpoint_state = staticmethod(_saver.update_checkpoint_state)
    checkpoint_exists = staticmethod(_saver.checkpoint_exists)
    Coordinator = _coord.Coordinator
    QueueRunner = _coord.QueueRunner
    add_queue_runner = staticmethod(_coord.add_queue_runner)
    start_queue_runners = staticmethod(_coord.start_queue_runners)
    batch = staticmethod(_input.batch)
    shuffle_batch = staticmethod(_input.shuffle_batch)
    input_producer = staticmethod(_input.input_producer)
    string_input_producer = staticmethod(_input.string_input_producer)
    range_input_producer = staticmethod(_input.range_input_producer)
    MonitoredSession = _ms.MonitoredSession
    MonitoredTrainingSession = staticmethod(_ms.MonitoredTrainingSession)
    Scaffold = _ms.Scaffold
    SessionManager = _ms.SessionManager
    ChiefSessionCreator = _ms.ChiefSessionCreator
    WorkerSessionCreator = _ms.WorkerSessionCreator
    SessionRunHook = _ms.SessionRunHook
    SessionRunArgs = _ms.SessionRunArgs
    SessionRunContext = _ms.SessionRunContext
    SessionRunValues = _ms.SessionRunValues
    StopAtStepHook = _ms.StopAtStepHook
    CheckpointSaverHook = _ms.CheckpointSaverHook
    LoggingTensorHook = _ms.LoggingTensorHook
    StepCounterHook = _ms.StepCounterHook
    NanTensorHook = _ms.NanTensorHook
    SummarySaverHook = _ms.SummarySaverHook
    Server = _server_lib.Server
    ClusterSpec = _server_lib.ClusterSpec
    replica_device_setter = staticmethod(_device_setter.replica_device_setter)


train = _TrainModule()


class _SummaryModule(object):
    scalar = staticmethod(_summary_mod.scalar)
    image = staticmethod(_summary_mod.image)
    audio = staticmethod(_summary_mod.audio)
    histogram = staticmethod(_summary_mod.histogram)
    merge = staticmethod(_summary_mod.merge)
    merge_all = staticmethod(_summary_mod.merge_all)
    FileWriter = _summary_writer.FileWriter


summary = _SummaryModule()


class _PythonIoModule(object):
    from simple_tensorflow_amd.python.lib.io.tf_record import (
        TFRecordWriter, tf_record_iterator)


python_io = _PythonIoModule()

from simple_tensorflow_amd.python import saved_model  # noqa: E402,F401
from simple_tensorflow_amd.python import estimator  # noqa: E402,F401
from simple_tensorflow_amd.python.ops import losses  # noqa: E402,F401
from simple_tensorflow_amd.python.ops import image_ops_impl as image  # noqa: E402,F401
from simple_tensorflow_amd.python.ops import sparse_ops as _sparse_ops  # noqa: E402
SparseTensor = _sparse_ops.SparseTensor
sparse_to_dense = _sparse_ops.sparse_to_dense
sparse_tensor_to_dense = _sparse_ops.sparse_tensor_to_dense
from simple_tensorflow_amd.python.layers import layers  # noqa: E402,F401
from simple_tensorflow_amd.python.platform import app  # noqa: E402,F401
from simple_tensorflow_amd.python.platform import gfile  # noqa: E402,F401
from simple_tensorflow_amd.python.platform import tf_logging as logging  # noqa: E402,F401
app.flags = __import__('simple_tensorflow_amd.python.platform.flags',
                       fromlist=['flags'])
flags = app.flags
from simple_tensorflow_amd.python.ops import metrics_impl as metrics  # noqa: E402,F401
from simple_tensorflow_amd.python.ops import gradient_checker as _gc  # noqa: E402
test = type(_sys)('simple_tensorflow_amd.test')
test.compute_gradient = _gc.compute_gradient
test.compute_gradient_error = _gc.compute_gradient_error
from simple_tensorflow_amd.python.client import timeline  # noqa: E402,F401
RunOptions = timeline.RunOptions
RunMetadata = timeline.RunMetadata

__version__ = '0.1.0'
