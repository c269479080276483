// Op registry definitions — the tf.* op surface (TF 1.0-compatible NodeDef
// attrs; reference: tensorflow/core/ops/*.cc). Grouped as in the reference's
// ops/ directory. Shape functions live in the Python layer
// (python/framework/shapes.py).
#include "framework/op.h"

namespace stf {

#define NUMTYPES "{float, double, int32, int64, bfloat16, half, uint8, int8, complex64}"
#define REALTYPES "{float, double, int32, int64, bfloat16, half}"
#define FLOATTYPES "{float, double, bfloat16, half}"

// ----------------------------- array_ops ----------------------------------
REGISTER_OP("Const").Output("output: dtype").Attr("dtype: type").Attr("value: tensor");
REGISTER_OP("Placeholder").Output("output: dtype").Attr("dtype: type").Attr("shape: shape = []");
REGISTER_OP("DepthwiseConv2dNative").Input("input: T").Input("filter: T").Output("output: T").Attr("T: {float, bfloat16}").Attr("strides: list(int)").Attr("padding: string").Attr("data_format: string = 'NHWC'");
REGISTER_OP("DepthwiseConv2dNativeBackpropInput").Input("input_sizes: int32").Input("filter: T").Input("out_backprop: T").Output("output: T").Attr("T: {float, bfloat16}").Attr("strides: list(int)").Attr("padding: string").Attr("data_format: string = 'NHWC'");
REGISTER_OP("DepthwiseConv2dNativeBackpropFilter").Input("input: T").Input("filter_sizes: int32").Input("out_backprop: T").Output("output: T").Attr("T: {float, bfloat16}").Attr("strides: list(int)").Attr("padding: string").Attr("data_format: string = 'NHWC'");
REGISTER_OP("PlaceholderWithDefault").Input("input: dtype").Output("output: dtype").Attr("dtype: type").Attr("shape: shape");
REGISTER_OP("Identity").Input("input: T").Output("output: T").Attr("T: type");
REGISTER_OP("StopGradient").Input("input: T").Output("output: T").Attr("T: type");
REGISTER_OP("PreventGradient").Input("input: T").Output("output: T").Attr("T: type").Attr("message: string = ''");
REGISTER_OP("Shape").Input("input: T").Output("output: out_type").Attr("T: type").Attr("out_type: {int32, int64} = int32");
REGISTER_OP("ShapeN").Input("input: N * T").Output("output: N * out_type").Attr("N: int >= 1").Attr("T: type").Attr("out_type: {int32, int64} = int32");
REGISTER_OP("Rank").Input("input: T").Output("output: int32").Attr("T: type");
REGISTER_OP("Size").Input("input: T").Output("output: out_type").Attr("T: type").Attr("out_type: {int32, int64} = int32");
REGISTER_OP("Reshape").Input("tensor: T").Input("shape: Tshape").Output("output: T").Attr("T: type").Attr("Tshape: {int32, int64} = int32");
REGISTER_OP("ExpandDims").Input("input: T").Input("dim: Tdim").Output("output: T").Attr("T: type").Attr("Tdim: {int32, int64} = int32");
REGISTER_OP("Squeeze").Input("input: T").Output("output: T").Attr("T: type").Attr("squeeze_dims: list(int) = []");
REGISTER_OP("Fill").Input("dims: int32").Input("value: T").Output("output: T").Attr("T: type");
REGISTER_OP("ZerosLike").Input("x: T").Output("y: T").Attr("T: type");
REGISTER_OP("OnesLike").Input("x: T").Output("y: T").Attr("T: type");
REGISTER_OP("Cast").Input("x: SrcT").Output("y: DstT").Attr("SrcT: type").Attr("DstT: type");
REGISTER_OP("Pack").Input("values: N * T").Output("output: T").Attr("N: int >= 1").Attr("T: type").Attr("axis: int = 0");
REGISTER_OP("Unpack").Input("value: T").Output("output: num * T").Attr("num: int >= 0").Attr("T: type").Attr("axis: int = 0");
REGISTER_OP("ConcatV2").Input("values: N * T").Input("axis: Tidx").Output("output: T").Attr("N: int >= 2").Attr("T: type").Attr("Tidx: {int32, int64} = int32");
REGISTER_OP("Concat").Input("concat_dim: int32").Input("values: N * T").Output("output: T").Attr("N: int >= 2").Attr("T: type");
REGISTER_OP("ConcatOffset").Input("concat_dim: int32").Input("shape: N * int32").Output("offset: N * int32").Attr("N: int >= 2");
REGISTER_OP("Split").Input("split_dim: int32").Input("value: T").Output("output: num_split * T").Attr("num_split: int >= 1").Attr("T: type");
REGISTER_OP("SplitV").Input("value: T").Input("size_splits: Tlen").Input("split_dim: int32").Output("output: num_split * T").Attr("num_split: int >= 1").Attr("T: type").Attr("Tlen: {int32, int64} = int64");
REGISTER_OP("Slice").Input("input: T").Input("begin: Index").Input("size: Index").Output("output: T").Attr("T: type").Attr("Index: {int32, int64}");
REGISTER_OP("StridedSlice").Input("input: T").Input("begin: Index").Input("end: Index").Input("strides: Index").Output("output: T").Attr("T: type").Attr("Index: {int32, int64}").Attr("begin_mask: int = 0").Attr("end_mask: int = 0").Attr("ellipsis_mask: int = 0").Attr("new_axis_mask: int = 0").Attr("shrink_axis_mask: int = 0");
REGISTER_OP("StridedSliceGrad").Input("shape: Index").Input("begin: Index").Input("end: Index").Input("strides: Index").Input("dy: T").Output("output: T").Attr("T: type").Attr("Index: {int32, int64}").Attr("begin_mask: int = 0").Attr("end_mask: int = 0").Attr("ellipsis_mask: int = 0").Attr("new_axis_mask: int = 0").Attr("shrink_axis_mask: int = 0");
REGISTER_OP("Pad").Input("input: T").Input("paddings: Tpaddings").Output("output: T").Attr("T: type").Attr("Tpaddings: {int32, int64} = int32");
REGISTER_OP("Transpose").Input("x: T").Input("perm: Tperm").Output("y: T").Attr("T: type").Attr("Tperm: {int32, int64} = int32");
REGISTER_OP("Gather").Input("params: Tparams").Input("indices: Tindices").Output("output: Tparams").Attr("validate_indices: bool = true").Attr("Tparams: type").Attr("Tindices: {int32, int64}");
REGISTER_OP("GatherV2").Input("params: Tparams").Input("indices: Tindices").Input("axis: Taxis").Output("output: Tparams").Attr("Tparams: type").Attr("Tindices: {int32, int64}").Attr("Taxis: {int32, int64} = int32");
REGISTER_OP("Tile").Input("input: T").Input("multiples: Tmultiples").Output("output: T").Attr("T: type").Attr("Tmultiples: {int32, int64} = int32");
REGISTER_OP("InvertPermutation").Input("x: T").Output("y: T").Attr("T: {int32, int64} = int32");
REGISTER_OP("Reverse").Input("tensor: T").Input("dims: bool").Output("output: T").Attr("T: type");
REGISTER_OP("BroadcastGradientArgs").Input("s0: T").Input("s1: T").Output("r0: T").Output("r1: T").Attr("T: {int32, int64} = int32");
REGISTER_OP("OneHot").Input("indices: TI").Input("depth: int32").Input("on_value: T").Input("off_value: T").Output("output: T").Attr("axis: int = -1").Attr("T: type").Attr("TI: {uint8, int32, int64} = int64");
REGISTER_OP("Range").Input("start: Tidx").Input("limit: Tidx").Input("delta: Tidx").Output("output: Tidx").Attr("Tidx: {float, double, int32, int64} = int32");
REGISTER_OP("LinSpace").Input("start: T").Input("stop: T").Input("num: Tidx").Output("output: T").Attr("T: {float, double}").Attr("Tidx: {int32, int64} = int32");
REGISTER_OP("CheckNumerics").Input("tensor: T").Output("output: T").Attr("T: " FLOATTYPES).Attr("message: string");
REGISTER_OP("UnsortedSegmentSum").Input("data: T").Input("segment_ids: Tindices").Input("num_segments: int32").Output("output: T").Attr("T: " NUMTYPES).Attr("Tindices: {int32, int64}");
REGISTER_OP("DynamicStitch").Input("indices: N * int32").Input("data: N * T").Output("merged: T").Attr("N: int >= 1").Attr("T: type");

// ----------------------------- math_ops -----------------------------------
#define BINARY_OP(NAME) REGISTER_OP(NAME).Input("x: T").Input("y: T").Output("z: T")
BINARY_OP("Add").Attr("T: " NUMTYPES);
BINARY_OP("Sub").Attr("T: " NUMTYPES);
BINARY_OP("Mul").Attr("T: " NUMTYPES);
BINARY_OP("Div").Attr("T: " NUMTYPES);
BINARY_OP("RealDiv").Attr("T: " NUMTYPES);
BINARY_OP("FloorDiv").Attr("T: " NUMTYPES);
BINARY_OP("FloorMod").Attr("T: " NUMTYPES);
BINARY_OP("Pow").Attr("T: " NUMTYPES);
BINARY_OP("Maximum").Attr("T: " REALTYPES);
BINARY_OP("Minimum").Attr("T: " REALTYPES);
BINARY_OP("SquaredDifference").Attr("T: " NUMTYPES);
#undef BINARY_OP
#define CMP_OP(NAME) REGISTER_OP(NAME).Input("x: T").Input("y: T").Output("z: bool")
CMP_OP("Less").Attr("T: " REALTYPES);
CMP_OP("LessEqual").Attr("T: " REALTYPES);
CMP_OP("Greater").Attr("T: " REALTYPES);
CMP_OP("GreaterEqual").Attr("T: " REALTYPES);
CMP_OP("Equal").Attr("T: " NUMTYPES);
CMP_OP("NotEqual").Attr("T: " NUMTYPES);
#undef CMP_OP
REGISTER_OP("LogicalAnd").Input("x: bool").Input("y: bool").Output("z: bool");
REGISTER_OP("LogicalOr").Input("x: bool").Input("y: bool").Output("z: bool");
REGISTER_OP("LogicalNot").Input("x: bool").Output("y: bool");
#define UNARY_OP(NAME) REGISTER_OP(NAME).Input("x: T").Output("y: T")
UNARY_OP("Neg").Attr("T: " NUMTYPES);
UNARY_OP("Abs").Attr("T: " REALTYPES);
UNARY_OP("Sign").Attr("T: " REALTYPES);
UNARY_OP("Square").Attr("T: " NUMTYPES);
UNARY_OP("Sqrt").Attr("T: " FLOATTYPES);
UNARY_OP("Rsqrt").Attr("T: " FLOATTYPES);
UNARY_OP("Exp").Attr("T: " FLOATTYPES);
UNARY_OP("Log").Attr("T: " FLOATTYPES);
UNARY_OP("Log1p").Attr("T: " FLOATTYPES);
UNARY_OP("Tanh").Attr("T: " FLOATTYPES);
UNARY_OP("Sigmoid").Attr("T: " FLOATTYPES);
UNARY_OP("Sin").Attr("T: " FLOATTYPES);
UNARY_OP("Cos").Attr("T: " FLOATTYPES);
UNARY_OP("Floor").Attr("T: " FLOATTYPES);
UNARY_OP("Ceil").Attr("T: " FLOATTYPES);
UNARY_OP("Round").Attr("T: " FLOATTYPES);
UNARY_OP("Reciprocal").Attr("T: " NUMTYPES);
#undef UNARY_OP
REGISTER_OP("IsNan").Input("x: T").Output("y: bool").Attr("T: " FLOATTYPES);
REGISTER_OP("IsInf").Input("x: T").Output("y: bool").Attr("T: " FLOATTYPES);
REGISTER_OP("IsFinite").Input("x: T").Output("y: bool").Attr("T: " FLOATTYPES);
#define GRAD2_OP(NAME) REGISTER_OP(NAME).Input("y: T").Input("dy: T").Output("z: T").Attr("T: " FLOATTYPES)
GRAD2_OP("SigmoidGrad");
GRAD2_OP("TanhGrad");
GRAD2_OP("RsqrtGrad");
GRAD2_OP("SqrtGrad");
GRAD2_OP("ReciprocalGrad");
#undef GRAD2_OP
REGISTER_OP("AddN").Input("inputs: N * T").Output("sum: T").Attr("N: int >= 1").Attr("T: " NUMTYPES);
REGISTER_OP("MatMul").Input("a: T").Input("b: T").Output("product: T").Attr("transpose_a: bool = false").Attr("transpose_b: bool = false").Attr("T: " FLOATTYPES);
REGISTER_OP("BatchMatMul").Input("x: T").Input("y: T").Output("output: T").Attr("T: " FLOATTYPES).Attr("adj_x: bool = false").Attr("adj_y: bool = false");
#define REDUCE_OP(NAME) REGISTER_OP(NAME).Input("input: T").Input("reduction_indices: Tidx").Output("output: T").Attr("keep_dims: bool = false").Attr("T: " NUMTYPES).Attr("Tidx: {int32, int64} = int32")
REDUCE_OP("Sum");
REDUCE_OP("Mean");
REDUCE_OP("Prod");
REDUCE_OP("Max");
REDUCE_OP("Min");
#undef REDUCE_OP
REGISTER_OP("All").Input("input: bool").Input("reduction_indices: Tidx").Output("output: bool").Attr("keep_dims: bool = false").Attr("Tidx: {int32, int64} = int32");
REGISTER_OP("Any").Input("input: bool").Input("reduction_indices: Tidx").Output("output: bool").Attr("keep_dims: bool = false").Attr("Tidx: {int32, int64} = int32");
REGISTER_OP("ArgMax").Input("input: T").Input("dimension: Tidx").Output("output: output_type").Attr("T: " NUMTYPES).Attr("Tidx: {int32, int64} = int32").Attr("output_type: {int32, int64} = int64");
REGISTER_OP("ArgMin").Input("input: T").Input("dimension: Tidx").Output("output: output_type").Attr("T: " NUMTYPES).Attr("Tidx: {int32, int64} = int32").Attr("output_type: {int32, int64} = int64");
REGISTER_OP("Select").Input("condition: bool").Input("t: T").Input("e: T").Output("output: T").Attr("T: type");
REGISTER_OP("Cumsum").Input("x: T").Input("axis: Tidx").Output("out: T").Attr("exclusive: bool = false").Attr("reverse: bool = false").Attr("T: " NUMTYPES).Attr("Tidx: {int32, int64} = int32");

// ------------------------------- nn_ops ------------------------------------
REGISTER_OP("Conv2D").Input("input: T").Input("filter: T").Output("output: T").Attr("T: " FLOATTYPES).Attr("strides: list(int)").Attr("use_cudnn_on_gpu: bool = true").Attr("padding: string").Attr("data_format: string = 'NHWC'");
REGISTER_OP("Conv2DBackpropInput").Input("input_sizes: int32").Input("filter: T").Input("out_backprop: T").Output("output: T").Attr("T: " FLOATTYPES).Attr("strides: list(int)").Attr("use_cudnn_on_gpu: bool = true").Attr("padding: string").Attr("data_format: string = 'NHWC'");
// Fused variant produced by gradient aggregation: output = conv-backprop-dx
// + side (residual-gradient accumulation folded into the GEMM epilogue).
REGISTER_OP("Conv2DBackpropInputAdd").Input("input_sizes: int32").Input("filter: T").Input("out_backprop: T").Input("side: T").Output("output: T").Attr("T: " FLOATTYPES).Attr("strides: list(int)").Attr("padding: string").Attr("data_format: string = 'NHWC'");
REGISTER_OP("Conv2DBackpropFilter").Input("input: T").Input("filter_sizes: int32").Input("out_backprop: T").Output("output: T").Attr("T: " FLOATTYPES).Attr("strides: list(int)").Attr("use_cudnn_on_gpu: bool = true").Attr("padding: string").Attr("data_format: string = 'NHWC'");
REGISTER_OP("BiasAdd").Input("value: T").Input("bias: T").Output("output: T").Attr("T: " NUMTYPES).Attr("data_format: string = 'NHWC'");
REGISTER_OP("BiasAddGrad").Input("out_backprop: T").Output("output: T").Attr("T: " NUMTYPES).Attr("data_format: string = 'NHWC'");
REGISTER_OP("Relu").Input("features: T").Output("activations: T").Attr("T: " REALTYPES);
REGISTER_OP("ReluGrad").Input("gradients: T").Input("features: T").Output("backprops: T").Attr("T: " REALTYPES);
REGISTER_OP("Relu6").Input("features: T").Output("activations: T").Attr("T: " REALTYPES);
REGISTER_OP("Relu6Grad").Input("gradients: T").Input("features: T").Output("backprops: T").Attr("T: " REALTYPES);
REGISTER_OP("Elu").Input("features: T").Output("activations: T").Attr("T: " FLOATTYPES);
REGISTER_OP("EluGrad").Input("gradients: T").Input("outputs: T").Output("backprops: T").Attr("T: " FLOATTYPES);
REGISTER_OP("Softplus").Input("features: T").Output("activations: T").Attr("T: " FLOATTYPES);
REGISTER_OP("SoftplusGrad").Input("gradients: T").Input("features: T").Output("backprops: T").Attr("T: " FLOATTYPES);
REGISTER_OP("Softmax").Input("logits: T").Output("softmax: T").Attr("T: " FLOATTYPES);
REGISTER_OP("LogSoftmax").Input("logits: T").Output("logsoftmax: T").Attr("T: " FLOATTYPES);
REGISTER_OP("SoftmaxCrossEntropyWithLogits").Input("features: T").Input("labels: T").Output("loss: T").Output("backprop: T").Attr("T: " FLOATTYPES);
REGISTER_OP("SparseSoftmaxCrossEntropyWithLogits").Input("features: T").Input("labels: Tlabels").Output("loss: T").Output("backprop: T").Attr("T: " FLOATTYPES).Attr("Tlabels: {int32, int64} = int64");
REGISTER_OP("MaxPool").Input("input: T").Output("output: T").Attr("T: " FLOATTYPES).Attr("ksize: list(int)").Attr("strides: list(int)").Attr("padding: string").Attr("data_format: string = 'NHWC'");
REGISTER_OP("MaxPoolGrad").Input("orig_input: T").Input("orig_output: T").Input("grad: T").Output("output: T").Attr("T: " FLOATTYPES).Attr("ksize: list(int)").Attr("strides: list(int)").Attr("padding: string").Attr("data_format: string = 'NHWC'");
REGISTER_OP("AvgPool").Input("value: T").Output("output: T").Attr("T: " FLOATTYPES).Attr("ksize: list(int)").Attr("strides: list(int)").Attr("padding: string").Attr("data_format: string = 'NHWC'");
REGISTER_OP("AvgPoolGrad").Input("orig_input_shape: int32").Input("grad: T").Output("output: T").Attr("T: " FLOATTYPES).Attr("ksize: list(int)").Attr("strides: list(int)").Attr("padding: string").Attr("data_format: string = 'NHWC'");
REGISTER_OP("FusedBatchNorm").Input("x: T").Input("scale: T").Input("offset: T").Input("mean: T").Input("variance: T").Output("y: T").Output("batch_mean: T").Output("batch_variance: T").Output("reserve_space_1: T").Output("reserve_space_2: T").Attr("T: {float}").Attr("epsilon: float = 0.0001").Attr("data_format: string = 'NHWC'").Attr("is_training: bool = true");
REGISTER_OP("FusedBatchNormGrad").Input("y_backprop: T").Input("x: T").Input("scale: T").Input("reserve_space_1: T").Input("reserve_space_2: T").Output("x_backprop: T").Output("scale_backprop: T").Output("offset_backprop: T").Output("reserve_space_3: T").Output("reserve_space_4: T").Attr("T: {float}").Attr("epsilon: float = 0.0001").Attr("data_format: string = 'NHWC'").Attr("is_training: bool = true");
// MI355X-native fused batch norm: bf16 activations with f32 statistics and
// f32 scale/offset (the bf16 analog of FusedBatchNorm; reserve = 1/sqrt(var+eps)).
REGISTER_OP("BatchNormMi").Input("x: T").Input("scale: float").Input("offset: float").Output("y: T").Output("batch_mean: float").Output("batch_variance: float").Output("reserve_inv_std: float").Attr("T: {float, bfloat16}").Attr("epsilon: float = 0.0001").Attr("fuse_relu: bool = false");
REGISTER_OP("BatchNormAddReluMi").Input("x: T").Input("scale: float").Input("offset: float").Input("side: T").Output("y: T").Output("batch_mean: float").Output("batch_variance: float").Output("reserve_inv_std: float").Attr("T: {float, bfloat16}").Attr("epsilon: float = 0.0001");
REGISTER_OP("BatchNormAddReluMiGrad").Input("y_backprop: T").Input("x: T").Input("scale: float").Input("saved_mean: float").Input("saved_inv_std: float").Input("y_relu: T").Output("x_backprop: T").Output("scale_backprop: float").Output("offset_backprop: float").Output("side_backprop: T").Attr("T: {float, bfloat16}").Attr("epsilon: float = 0.0001");
REGISTER_OP("BatchNormMiGrad").Input("y_backprop: T").Input("x: T").Input("scale: float").Input("saved_mean: float").Input("saved_inv_std: float").Input("y_relu: T").Output("x_backprop: T").Output("scale_backprop: float").Output("offset_backprop: float").Attr("T: {float, bfloat16}").Attr("epsilon: float = 0.0001").Attr("fuse_relu: bool = false");
// Internal: scoped elementwise fusion output (graph/optimizer.cc emits it
// post-autodiff; never constructed from python). Input 0 is the root
// tensor, inputs 1..N-1 are scalar side-inputs; `program` is the packed
// bytecode of kernels/fused_ew.h.
REGISTER_OP("_FusedElementwise").Input("inputs: N * T").Output("output: T").Attr("N: int >= 1").Attr("T: {float, bfloat16}").Attr("program: list(int)");
REGISTER_OP("L2Loss").Input("t: T").Output("output: T").Attr("T: " FLOATTYPES);
// Fused LSTM cell pointwise math over the post-GEMM gate matrix [B, 4H]
// (gate order i, j, f, o — BasicLSTMCell split order). Outputs every
// activation the backward op needs (reference capability analog:
// contrib/rnn lstm_ops.cc LSTMBlockCell / LSTMBlockCellGrad).
REGISTER_OP("LSTMGates").Input("gates: T").Input("c_prev: T").Output("i: T").Output("f: T").Output("o: T").Output("ci: T").Output("cs: T").Output("co: T").Output("h: T").Attr("T: {float, bfloat16}").Attr("forget_bias: float = 1.0");
REGISTER_OP("LSTMGatesGrad").Input("c_prev: T").Input("i: T").Input("f: T").Input("o: T").Input("ci: T").Input("co: T").Input("dh: T").Input("dcs: T").Output("dgates: T").Output("dc_prev: T").Attr("T: {float, bfloat16}");
// image ops (reference core/ops/image_ops.cc; kernels/cpu_image.cc)
REGISTER_OP("RGBToHSV").Input("images: T").Output("output: T").Attr("T: {float, double} = float");
REGISTER_OP("HSVToRGB").Input("images: T").Output("output: T").Attr("T: {float, double} = float");
REGISTER_OP("AdjustContrastv2").Input("images: T").Input("contrast_factor: float").Output("output: T").Attr("T: {float} = float");
REGISTER_OP("NonMaxSuppression").Input("boxes: float").Input("scores: float").Input("max_output_size: int32").Output("selected_indices: int32").Attr("iou_threshold: float = 0.5");
REGISTER_OP("NonMaxSuppressionV2").Input("boxes: float").Input("scores: float").Input("max_output_size: int32").Input("iou_threshold: float").Output("selected_indices: int32");
REGISTER_OP("SampleDistortedBoundingBox").Input("image_size: T").Input("bounding_boxes: float").Output("begin: T").Output("size: T").Output("bboxes: float").Attr("T: {int32, int64} = int32").Attr("seed: int = 0").Attr("seed2: int = 0").Attr("min_object_covered: float = 0.1").Attr("aspect_ratio_range: list(float) = [0.75, 1.33]").Attr("area_range: list(float) = [0.05, 1.0]").Attr("max_attempts: int = 100").Attr("use_image_if_no_bounding_boxes: bool = false").SetIsStateful();
REGISTER_OP("LRN").Input("input: T").Output("output: T").Attr("depth_radius: int = 5").Attr("bias: float = 1.0").Attr("alpha: float = 1.0").Attr("beta: float = 0.5").Attr("T: {float}");
REGISTER_OP("InTopK").Input("predictions: float").Input("targets: T").Output("precision: bool").Attr("k: int").Attr("T: {int32, int64} = int32");

// ------------------------- candidate sampling ------------------------------
// Reference core/ops/candidate_sampling_ops.cc; kernels in
// kernels/cpu_sampling.cc. true_classes is [batch, num_true] int64.
#define SAMPLER_OP(NAME) REGISTER_OP(NAME).Input("true_classes: int64").Output("sampled_candidates: int64").Output("true_expected_count: float").Output("sampled_expected_count: float").Attr("num_true: int >= 1").Attr("num_sampled: int >= 1").Attr("unique: bool").Attr("range_max: int >= 1").Attr("seed: int = 0").Attr("seed2: int = 0").SetIsStateful()
SAMPLER_OP("UniformCandidateSampler");
SAMPLER_OP("LogUniformCandidateSampler");
SAMPLER_OP("LearnedUnigramCandidateSampler");
#undef SAMPLER_OP
REGISTER_OP("ComputeAccidentalHits").Input("true_classes: int64").Input("sampled_candidates: int64").Output("indices: int32").Output("ids: int64").Output("weights: float").Attr("num_true: int").Attr("seed: int = 0").Attr("seed2: int = 0");

// --------------------------------- CTC -------------------------------------
// Reference core/ops/ctc_ops.cc / kernels/ctc_loss_op.cc. Kernel in
// kernels/cpu_ctc.cc (log-space forward-backward).
REGISTER_OP("CTCLoss").Input("inputs: float").Input("labels_indices: int64").Input("labels_values: int32").Input("sequence_length: int32").Output("loss: float").Output("gradient: float").Attr("preprocess_collapse_repeated: bool = false").Attr("ctc_merge_repeated: bool = true");
REGISTER_OP("CTCGreedyDecoder").Input("inputs: float").Input("sequence_length: int32").Output("decoded_indices: int64").Output("decoded_values: int64").Output("decoded_shape: int64").Output("log_probability: float").Attr("merge_repeated: bool = false");

// ----------------------------- spectral (FFT) ------------------------------
// Reference core/ops/spectral_ops.cc. CPU kernels in kernels/cpu_fft.cc
// (iterative radix-2 + Bluestein chirp-z for arbitrary lengths, double
// accumulation).
REGISTER_OP("FFT").Input("input: complex64").Output("output: complex64");
REGISTER_OP("IFFT").Input("input: complex64").Output("output: complex64");
REGISTER_OP("FFT2D").Input("input: complex64").Output("output: complex64");
REGISTER_OP("IFFT2D").Input("input: complex64").Output("output: complex64");
REGISTER_OP("FFT3D").Input("input: complex64").Output("output: complex64");
REGISTER_OP("IFFT3D").Input("input: complex64").Output("output: complex64");
REGISTER_OP("RFFT").Input("input: float").Input("fft_length: int32").Output("output: complex64");
REGISTER_OP("IRFFT").Input("input: complex64").Input("fft_length: int32").Output("output: float");
REGISTER_OP("RFFT2D").Input("input: float").Input("fft_length: int32").Output("output: complex64");
REGISTER_OP("IRFFT2D").Input("input: complex64").Input("fft_length: int32").Output("output: float");
// complex construction/accessors (reference math_ops.cc)
REGISTER_OP("Complex").Input("real: T").Input("imag: T").Output("out: Tout").Attr("T: {float, double} = float").Attr("Tout: {complex64, complex128} = complex64");
REGISTER_OP("Real").Input("input: T").Output("output: Tout").Attr("T: {complex64, complex128} = complex64").Attr("Tout: {float, double} = float");
REGISTER_OP("Imag").Input("input: T").Output("output: Tout").Attr("T: {complex64, complex128} = complex64").Attr("Tout: {float, double} = float");
REGISTER_OP("Conj").Input("input: T").Output("output: T").Attr("T: {complex64, complex128} = complex64");
REGISTER_OP("ComplexAbs").Input("x: T").Output("y: Tout").Attr("T: {complex64, complex128} = complex64").Attr("Tout: {float, double} = float");

// -------------------------------- set ops ----------------------------------
// Reference core/ops/set_ops.cc (dense-to-dense subset; sparse outputs).
REGISTER_OP("DenseToDenseSetOperation").Input("set1: T").Input("set2: T").Output("result_indices: int64").Output("result_values: T").Output("result_shape: int64").Attr("set_operation: string").Attr("validate_indices: bool = true").Attr("T: {int32, int64}");
REGISTER_OP("SetSize").Input("set_indices: int64").Input("set_values: T").Input("set_shape: int64").Output("size: int32").Attr("validate_indices: bool = true").Attr("T: {int32, int64}");

// ------------------------------ quantized ----------------------------------
// Reference core/ops/math_ops.cc quantized section + quantize_op.cc /
// quantized_matmul_op.cc. Carrier-typed redesign: quint8 rides DT_UINT8 and
// qint32 rides DT_INT32 (this framework has no distinct quantized dtypes;
// the range tensors carry the scale exactly as in the reference).
REGISTER_OP("QuantizeV2").Input("input: float").Input("min_range: float").Input("max_range: float").Output("output: T").Output("output_min: float").Output("output_max: float").Attr("T: {uint8} = uint8").Attr("mode: string = 'MIN_COMBINED'");
REGISTER_OP("Dequantize").Input("input: T").Input("min_range: float").Input("max_range: float").Output("output: float").Attr("T: {uint8, int32} = uint8").Attr("mode: string = 'MIN_COMBINED'");
REGISTER_OP("QuantizedMatMul").Input("a: T1").Input("b: T2").Input("min_a: float").Input("max_a: float").Input("min_b: float").Input("max_b: float").Output("out: Toutput").Output("min_out: float").Output("max_out: float").Attr("T1: {uint8} = uint8").Attr("T2: {uint8} = uint8").Attr("Toutput: {int32} = int32").Attr("transpose_a: bool = false").Attr("transpose_b: bool = false");
REGISTER_OP("QuantizedRelu").Input("features: T").Input("min_features: float").Input("max_features: float").Output("activations: T").Output("min_activations: float").Output("max_activations: float").Attr("T: {uint8} = uint8");
REGISTER_OP("QuantizeDownAndShrinkRange").Input("input: Tinput").Input("input_min: float").Input("input_max: float").Output("output: out_type").Output("output_min: float").Output("output_max: float").Attr("Tinput: {int32} = int32").Attr("out_type: {uint8} = uint8");
REGISTER_OP("RequantizationRange").Input("input: Tinput").Input("input_min: float").Input("input_max: float").Output("output_min: float").Output("output_max: float").Attr("Tinput: {int32} = int32");

// --------------------------- linear algebra --------------------------------
// Batched dense decompositions (reference core/ops/linalg_ops.cc). CPU
// kernels in kernels/cpu_linalg.cc (LU w/ partial pivoting, Cholesky,
// Householder QR, one-sided Jacobi SVD, cyclic Jacobi symmetric eig).
REGISTER_OP("Cholesky").Input("input: T").Output("output: T").Attr("T: {float, double}");
REGISTER_OP("CholeskyGrad").Input("l: T").Input("grad: T").Output("output: T").Attr("T: {float, double}");
REGISTER_OP("MatrixDeterminant").Input("input: T").Output("output: T").Attr("T: {float, double}");
REGISTER_OP("MatrixInverse").Input("input: T").Output("output: T").Attr("adjoint: bool = false").Attr("T: {float, double}");
REGISTER_OP("MatrixSolve").Input("matrix: T").Input("rhs: T").Output("output: T").Attr("adjoint: bool = false").Attr("T: {float, double}");
REGISTER_OP("MatrixTriangularSolve").Input("matrix: T").Input("rhs: T").Output("output: T").Attr("lower: bool = true").Attr("adjoint: bool = false").Attr("T: {float, double}");
REGISTER_OP("MatrixSolveLs").Input("matrix: T").Input("rhs: T").Input("l2_regularizer: double").Output("output: T").Attr("T: {float, double}").Attr("fast: bool = true");
REGISTER_OP("Qr").Input("input: T").Output("q: T").Output("r: T").Attr("full_matrices: bool = false").Attr("T: {float, double}");
REGISTER_OP("Svd").Input("input: T").Output("s: T").Output("u: T").Output("v: T").Attr("compute_uv: bool = true").Attr("full_matrices: bool = false").Attr("T: {float, double}");
REGISTER_OP("SelfAdjointEigV2").Input("input: T").Output("e: T").Output("v: T").Attr("compute_v: bool = true").Attr("T: {float, double}");

// --------------------------- state / training -------------------------------
REGISTER_OP("VariableV2").Output("ref: Ref(dtype)").Attr("shape: shape").Attr("dtype: type").Attr("container: string = ''").Attr("shared_name: string = ''").SetIsStateful();
REGISTER_OP("Variable").Output("ref: Ref(dtype)").Attr("shape: shape").Attr("dtype: type").Attr("container: string = ''").Attr("shared_name: string = ''").SetIsStateful();
REGISTER_OP("IsVariableInitialized").Input("ref: Ref(dtype)").Output("is_initialized: bool").Attr("dtype: type").SetAllowsUninitializedInput();
REGISTER_OP("Assign").Input("ref: Ref(T)").Input("value: T").Output("output_ref: Ref(T)").Attr("T: type").Attr("validate_shape: bool = true").Attr("use_locking: bool = true").SetAllowsUninitializedInput();
REGISTER_OP("AssignAdd").Input("ref: Ref(T)").Input("value: T").Output("output_ref: Ref(T)").Attr("T: " NUMTYPES).Attr("use_locking: bool = false");
REGISTER_OP("AssignSub").Input("ref: Ref(T)").Input("value: T").Output("output_ref: Ref(T)").Attr("T: " NUMTYPES).Attr("use_locking: bool = false");
REGISTER_OP("ScatterSub").Input("ref: Ref(T)").Input("indices: Tindices").Input("updates: T").Output("output_ref: Ref(T)").Attr("T: " NUMTYPES).Attr("Tindices: {int32, int64}").Attr("use_locking: bool = false");
REGISTER_OP("ScatterAdd").Input("ref: Ref(T)").Input("indices: Tindices").Input("updates: T").Output("output_ref: Ref(T)").Attr("T: " NUMTYPES).Attr("Tindices: {int32, int64}").Attr("use_locking: bool = false");
REGISTER_OP("ApplyGradientDescent").Input("var: Ref(T)").Input("alpha: T").Input("delta: T").Output("out: Ref(T)").Attr("T: " FLOATTYPES).Attr("use_locking: bool = false");
REGISTER_OP("ApplyMomentum").Input("var: Ref(T)").Input("accum: Ref(T)").Input("lr: T").Input("grad: T").Input("momentum: T").Output("out: Ref(T)").Attr("T: " FLOATTYPES).Attr("use_locking: bool = false").Attr("use_nesterov: bool = false");
REGISTER_OP("ApplyAdam").Input("var: Ref(T)").Input("m: Ref(T)").Input("v: Ref(T)").Input("beta1_power: T").Input("beta2_power: T").Input("lr: T").Input("beta1: T").Input("beta2: T").Input("epsilon: T").Input("grad: T").Output("out: Ref(T)").Attr("T: " FLOATTYPES).Attr("use_locking: bool = false");
REGISTER_OP("ApplyRMSProp").Input("var: Ref(T)").Input("ms: Ref(T)").Input("mom: Ref(T)").Input("lr: T").Input("rho: T").Input("momentum: T").Input("epsilon: T").Input("grad: T").Output("out: Ref(T)").Attr("T: " FLOATTYPES).Attr("use_locking: bool = false");
REGISTER_OP("ApplyAdagrad").Input("var: Ref(T)").Input("accum: Ref(T)").Input("lr: T").Input("grad: T").Output("out: Ref(T)").Attr("T: " FLOATTYPES).Attr("use_locking: bool = false");
REGISTER_OP("ApplyAdadelta").Input("var: Ref(T)").Input("accum: Ref(T)").Input("accum_update: Ref(T)").Input("lr: T").Input("rho: T").Input("epsilon: T").Input("grad: T").Output("out: Ref(T)").Attr("T: " FLOATTYPES).Attr("use_locking: bool = false");
REGISTER_OP("ApplyFtrl").Input("var: Ref(T)").Input("accum: Ref(T)").Input("linear: Ref(T)").Input("grad: T").Input("lr: T").Input("l1: T").Input("l2: T").Input("lr_power: T").Output("out: Ref(T)").Attr("T: " FLOATTYPES).Attr("use_locking: bool = false");
REGISTER_OP("ApplyProximalGradientDescent").Input("var: Ref(T)").Input("alpha: T").Input("l1: T").Input("l2: T").Input("delta: T").Output("out: Ref(T)").Attr("T: " FLOATTYPES).Attr("use_locking: bool = false");
REGISTER_OP("CountUpTo").Input("ref: Ref(T)").Output("output: T").Attr("limit: int").Attr("T: {int32, int64}");
REGISTER_OP("DestroyTemporaryVariable").Input("ref: Ref(T)").Output("value: T").Attr("T: type").Attr("var_name: string");

// ------------------------------ random_ops ---------------------------------
REGISTER_OP("RandomUniform").Input("shape: T").Output("output: dtype").Attr("seed: int = 0").Attr("seed2: int = 0").Attr("dtype: " FLOATTYPES).Attr("T: {int32, int64}").SetIsStateful();
REGISTER_OP("RandomStandardNormal").Input("shape: T").Output("output: dtype").Attr("seed: int = 0").Attr("seed2: int = 0").Attr("dtype: " FLOATTYPES).Attr("T: {int32, int64}").SetIsStateful();
REGISTER_OP("TruncatedNormal").Input("shape: T").Output("output: dtype").Attr("seed: int = 0").Attr("seed2: int = 0").Attr("dtype: " FLOATTYPES).Attr("T: {int32, int64}").SetIsStateful();
REGISTER_OP("RandomUniformInt").Input("shape: T").Input("minval: Tout").Input("maxval: Tout").Output("output: Tout").Attr("seed: int = 0").Attr("seed2: int = 0").Attr("T: {int32, int64}").Attr("Tout: {int32, int64}").SetIsStateful();
REGISTER_OP("RandomShuffle").Input("value: T").Output("output: T").Attr("seed: int = 0").Attr("seed2: int = 0").Attr("T: type").SetIsStateful();
REGISTER_OP("Multinomial").Input("logits: T").Input("num_samples: int32").Output("output: int64").Attr("seed: int = 0").Attr("seed2: int = 0").Attr("T: " REALTYPES).SetIsStateful();

// ----------------------------- control flow --------------------------------
REGISTER_OP("Switch").Input("data: T").Input("pred: bool").Output("output_false: T").Output("output_true: T").Attr("T: type");
REGISTER_OP("RefSwitch").Input("data: Ref(T)").Input("pred: bool").Output("output_false: Ref(T)").Output("output_true: Ref(T)").Attr("T: type").SetAllowsUninitializedInput();
REGISTER_OP("Merge").Input("inputs: N * T").Output("output: T").Output("value_index: int32").Attr("N: int >= 1").Attr("T: type");
REGISTER_OP("RefMerge").Input("inputs: N * Ref(T)").Output("output: Ref(T)").Output("value_index: int32").Attr("N: int >= 1").Attr("T: type");
REGISTER_OP("Enter").Input("data: T").Output("output: T").Attr("T: type").Attr("frame_name: string").Attr("is_constant: bool = false").Attr("parallel_iterations: int = 10");
REGISTER_OP("RefEnter").Input("data: Ref(T)").Output("output: Ref(T)").Attr("T: type").Attr("frame_name: string").Attr("is_constant: bool = false").Attr("parallel_iterations: int = 10");
REGISTER_OP("Exit").Input("data: T").Output("output: T").Attr("T: type");
REGISTER_OP("RefExit").Input("data: Ref(T)").Output("output: Ref(T)").Attr("T: type");
REGISTER_OP("NextIteration").Input("data: T").Output("output: T").Attr("T: type");
REGISTER_OP("RefNextIteration").Input("data: Ref(T)").Output("output: Ref(T)").Attr("T: type");
REGISTER_OP("LoopCond").Input("input: bool").Output("output: bool");
REGISTER_OP("ControlTrigger");
REGISTER_OP("NoOp");

// ------------------------------ send/recv ----------------------------------
REGISTER_OP("_Send").Input("tensor: T").Attr("T: type").Attr("tensor_name: string").Attr("send_device: string").Attr("send_device_incarnation: int = 0").Attr("recv_device: string").Attr("client_terminated: bool = false").SetIsStateful();
REGISTER_OP("_Recv").Output("tensor: tensor_type").Attr("tensor_type: type").Attr("tensor_name: string").Attr("send_device: string").Attr("send_device_incarnation: int = 0").Attr("recv_device: string").Attr("client_terminated: bool = false").SetIsStateful();

// ------------------------------ logging/debug -------------------------------
REGISTER_OP("Assert").Input("condition: bool").Input("data: T").Attr("T: list(type)").Attr("summarize: int = 3").SetIsStateful();
REGISTER_OP("Print").Input("input: T").Input("data: U").Output("output: T").Attr("T: type").Attr("U: list(type)").Attr("message: string = ''").Attr("first_n: int = -1").Attr("summarize: int = 3").SetIsStateful();
REGISTER_OP("ScalarSummary").Input("tags: string").Input("values: T").Output("summary: string").Attr("T: " REALTYPES);
REGISTER_OP("HistogramSummary").Input("tag: string").Input("values: T").Output("summary: string").Attr("T: " REALTYPES);
REGISTER_OP("MergeSummary").Input("inputs: N * string").Output("summary: string").Attr("N: int >= 1");

// ------------------------------ io / ckpt ----------------------------------
REGISTER_OP("SaveV2").Input("prefix: string").Input("tensor_names: string").Input("shape_and_slices: string").Input("tensors: dtypes").Attr("dtypes: list(type)").SetIsStateful();
REGISTER_OP("RestoreV2").Input("prefix: string").Input("tensor_names: string").Input("shape_and_slices: string").Output("tensors: dtypes").Attr("dtypes: list(type)").SetIsStateful();
REGISTER_OP("MergeV2Checkpoints").Input("checkpoint_prefixes: string").Input("destination_prefix: string").Attr("delete_old_dirs: bool = true").SetIsStateful();

// ------------------------------- queues ------------------------------------
// TensorArray family (reference core/ops/data_flow_ops.cc:1080; resource
// handles are session-scoped strings here, like the queue ops above).
REGISTER_OP("PlaceholderWithDefault").Input("input: dtype").Output("output: dtype").Attr("dtype: type").Attr("shape: shape = []");
REGISTER_OP("SparseToDense").Input("sparse_indices: Tindices").Input("output_shape: Tindices").Input("sparse_values: T").Input("default_value: T").Output("dense: T").Attr("validate_indices: bool = true").Attr("T: type").Attr("Tindices: {int32, int64}");

// sparse algebra (reference core/ops/sparse_ops.cc; kernels/cpu_sparse.cc)
REGISTER_OP("SparseAdd").Input("a_indices: int64").Input("a_values: T").Input("a_shape: int64").Input("b_indices: int64").Input("b_values: T").Input("b_shape: int64").Input("thresh: Treal").Output("sum_indices: int64").Output("sum_values: T").Output("sum_shape: int64").Attr("T: " NUMTYPES).Attr("Treal: {float, double, int32, int64}");
REGISTER_OP("SparseTensorDenseAdd").Input("a_indices: Tindices").Input("a_values: T").Input("a_shape: Tindices").Input("b: T").Output("output: T").Attr("T: " NUMTYPES).Attr("Tindices: {int32, int64}");
REGISTER_OP("SparseReorder").Input("input_indices: int64").Input("input_values: T").Input("input_shape: int64").Output("output_indices: int64").Output("output_values: T").Attr("T: type");
REGISTER_OP("SparseReduceSum").Input("input_indices: int64").Input("input_values: T").Input("input_shape: int64").Input("reduction_axes: int32").Output("output: T").Attr("keep_dims: bool = false").Attr("T: " NUMTYPES);
REGISTER_OP("SparseConcat").Input("indices: N * int64").Input("values: N * T").Input("shapes: N * int64").Output("output_indices: int64").Output("output_values: T").Output("output_shape: int64").Attr("concat_dim: int").Attr("N: int >= 2").Attr("T: type");REGISTER_OP("ResizeBilinear").Input("images: T").Input("size: int32").Output("resized_images: float").Attr("T: {float}").Attr("align_corners: bool = false");
REGISTER_OP("ResizeBilinearGrad").Input("grads: float").Input("original_image: T").Output("output: T").Attr("T: {float}").Attr("align_corners: bool = false");
REGISTER_OP("ResizeNearestNeighbor").Input("images: T").Input("size: int32").Output("resized_images: T").Attr("T: {float}").Attr("align_corners: bool = false");
REGISTER_OP("ConditionalAccumulator").Output("handle: Ref(string)").Attr("dtype: type").Attr("shape: shape").Attr("container: string = ''").Attr("shared_name: string = ''").SetIsStateful();
REGISTER_OP("AccumulatorApplyGradient").Input("handle: Ref(string)").Input("local_step: int64").Input("gradient: dtype").Attr("dtype: type").SetIsStateful();
REGISTER_OP("AccumulatorTakeGradient").Input("handle: Ref(string)").Input("num_required: int32").Output("average: dtype").Attr("dtype: type").SetIsStateful();
REGISTER_OP("AccumulatorSetGlobalStep").Input("handle: Ref(string)").Input("new_global_step: int64").SetIsStateful();
REGISTER_OP("AccumulatorNumAccumulated").Input("handle: Ref(string)").Output("num_accumulated: int32").SetIsStateful();
REGISTER_OP("PyFunc").Input("input: Tin").Output("output: Tout").Attr("token: string").Attr("Tin: list(type) >= 0").Attr("Tout: list(type) >= 0").SetIsStateful();
REGISTER_OP("PyFuncStateless").Input("input: Tin").Output("output: Tout").Attr("token: string").Attr("Tin: list(type) >= 0").Attr("Tout: list(type) >= 0");
REGISTER_OP("TensorArrayV3").Input("size: int32").Output("handle: string").Output("flow: float").Attr("dtype: type").Attr("dynamic_size: bool = false").Attr("clear_after_read: bool = true").Attr("tensor_array_name: string = ''").SetIsStateful();
REGISTER_OP("TensorArrayGradV3").Input("handle: string").Input("flow_in: float").Output("grad_handle: string").Output("flow_out: float").Attr("source: string").SetIsStateful();
REGISTER_OP("TensorArrayWriteV3").Input("handle: string").Input("index: int32").Input("value: T").Input("flow_in: float").Output("flow_out: float").Attr("T: type").SetIsStateful();
REGISTER_OP("TensorArrayReadV3").Input("handle: string").Input("index: int32").Input("flow_in: float").Output("value: dtype").Attr("dtype: type").SetIsStateful();
REGISTER_OP("TensorArraySizeV3").Input("handle: string").Input("flow_in: float").Output("size: int32").SetIsStateful();
REGISTER_OP("TensorArrayGatherV3").Input("handle: string").Input("indices: int32").Input("flow_in: float").Output("value: dtype").Attr("dtype: type").SetIsStateful();
REGISTER_OP("TensorArrayScatterV3").Input("handle: string").Input("indices: int32").Input("value: T").Input("flow_in: float").Output("flow_out: float").Attr("T: type").SetIsStateful();
REGISTER_OP("TensorArrayCloseV3").Input("handle: string").SetIsStateful();

REGISTER_OP("FIFOQueue").Output("handle: Ref(string)").Attr("component_types: list(type)").Attr("shapes: list(shape) = []").Attr("capacity: int = -1").Attr("container: string = ''").Attr("shared_name: string = ''").SetIsStateful();
REGISTER_OP("RandomShuffleQueue").Output("handle: Ref(string)").Attr("component_types: list(type)").Attr("shapes: list(shape) = []").Attr("capacity: int = -1").Attr("min_after_dequeue: int = 0").Attr("seed: int = 0").Attr("seed2: int = 0").Attr("container: string = ''").Attr("shared_name: string = ''").SetIsStateful();
REGISTER_OP("PaddingFIFOQueue").Output("handle: Ref(string)").Attr("component_types: list(type)").Attr("shapes: list(shape) = []").Attr("capacity: int = -1").Attr("container: string = ''").Attr("shared_name: string = ''").SetIsStateful();
REGISTER_OP("QueueEnqueue").Input("handle: Ref(string)").Input("components: Tcomponents").Attr("Tcomponents: list(type)").Attr("timeout_ms: int = -1").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("QueueEnqueueMany").Input("handle: Ref(string)").Input("components: Tcomponents").Attr("Tcomponents: list(type)").Attr("timeout_ms: int = -1").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("QueueDequeue").Input("handle: Ref(string)").Output("components: component_types").Attr("component_types: list(type)").Attr("timeout_ms: int = -1").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("QueueDequeueMany").Input("handle: Ref(string)").Input("n: int32").Output("components: component_types").Attr("component_types: list(type)").Attr("timeout_ms: int = -1").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("QueueClose").Input("handle: Ref(string)").Attr("cancel_pending_enqueues: bool = false").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("QueueSize").Input("handle: Ref(string)").Output("size: int32").SetIsStateful().SetAllowsUninitializedInput();

// ----------------------------- collectives ---------------------------------
// MI355X-native: RCCL collectives over xGMI as first-class graph ops
// (the reference had no collectives — §2.3 of SURVEY.md; gradient
// aggregation there was AddN/PS. Here all-reduce is the primary multi-GPU
// path per BASELINE.json config 3).
REGISTER_OP("RcclAllReduce").Input("input: T").Output("output: T").Attr("T: {float, bfloat16, half}").Attr("reduction: string = 'sum'").Attr("num_devices: int = 1").Attr("shared_name: string = ''").SetIsStateful();
REGISTER_OP("RcclBroadcast").Input("input: T").Output("output: T").Attr("T: {float, bfloat16, half}").Attr("root: int = 0").SetIsStateful();
// Fused gradient bucket: one ncclAllReduce per bucket on a dedicated comm
// stream (pack -> reduce f32 -> unpack*scale), overlapped with backprop.
// Outputs are valid for compute-stream consumers only AFTER a RcclCommSync
// control edge (parallel/dist.py wires it).
REGISTER_OP("RcclBucketAllReduce").Input("inputs: N * T").Output("outputs: N * T").Attr("N: int >= 1").Attr("T: {float, bfloat16}").Attr("scale: float = 1").SetIsStateful();
REGISTER_OP("RcclCommSync").SetIsStateful();

// ------------------------ breadth wave (round 2) ---------------------------
// search / index
REGISTER_OP("Where").Input("input: bool").Output("index: int64");
REGISTER_OP("Unique").Input("x: T").Output("y: T").Output("idx: out_idx").Attr("T: type").Attr("out_idx: {int32, int64} = int32");
REGISTER_OP("UniqueWithCounts").Input("x: T").Output("y: T").Output("idx: out_idx").Output("count: out_idx").Attr("T: type").Attr("out_idx: {int32, int64} = int32");
REGISTER_OP("TopKV2").Input("input: T").Input("k: int32").Output("values: T").Output("indices: int32").Attr("sorted: bool = true").Attr("T: " REALTYPES);
REGISTER_OP("TopK").Input("input: T").Output("values: T").Output("indices: int32").Attr("k: int >= 0").Attr("sorted: bool = true").Attr("T: " REALTYPES);
// scan / segments
REGISTER_OP("Cumprod").Input("x: T").Input("axis: Tidx").Output("out: T").Attr("exclusive: bool = false").Attr("reverse: bool = false").Attr("T: " NUMTYPES).Attr("Tidx: {int32, int64} = int32");
#define SEGMENT_OP(NAME) REGISTER_OP(NAME).Input("data: T").Input("segment_ids: Tindices").Output("output: T").Attr("T: " NUMTYPES).Attr("Tindices: {int32, int64}")
SEGMENT_OP("SegmentSum");
SEGMENT_OP("SegmentMean");
SEGMENT_OP("SegmentMax");
SEGMENT_OP("SegmentMin");
SEGMENT_OP("SegmentProd");
#undef SEGMENT_OP
// array restructuring
REGISTER_OP("ReverseV2").Input("tensor: T").Input("axis: Tidx").Output("output: T").Attr("T: type").Attr("Tidx: {int32, int64} = int32");
REGISTER_OP("ListDiff").Input("x: T").Input("y: T").Output("out: T").Output("idx: out_idx").Attr("T: type").Attr("out_idx: {int32, int64} = int32");
REGISTER_OP("DynamicPartition").Input("data: T").Input("partitions: int32").Output("outputs: num_partitions * T").Attr("num_partitions: int >= 1").Attr("T: type");
REGISTER_OP("GatherNd").Input("params: Tparams").Input("indices: Tindices").Output("output: Tparams").Attr("Tparams: type").Attr("Tindices: {int32, int64}");
REGISTER_OP("ScatterNd").Input("indices: Tindices").Input("updates: T").Input("shape: Tindices").Output("output: T").Attr("T: " NUMTYPES).Attr("Tindices: {int32, int64}");
REGISTER_OP("Diag").Input("diagonal: T").Output("output: T").Attr("T: " NUMTYPES);
REGISTER_OP("DiagPart").Input("input: T").Output("diagonal: T").Attr("T: " NUMTYPES);
REGISTER_OP("MatrixDiag").Input("diagonal: T").Output("output: T").Attr("T: type");
REGISTER_OP("BatchMatrixDiag").Input("diagonal: T").Output("output: T").Attr("T: type");
REGISTER_OP("MatrixDiagPart").Input("input: T").Output("diagonal: T").Attr("T: type");
REGISTER_OP("BatchMatrixDiagPart").Input("input: T").Output("diagonal: T").Attr("T: type");
REGISTER_OP("MatrixSetDiag").Input("input: T").Input("diagonal: T").Output("output: T").Attr("T: type");
REGISTER_OP("BatchMatrixSetDiag").Input("input: T").Input("diagonal: T").Output("output: T").Attr("T: type");
REGISTER_OP("MatrixBandPart").Input("input: T").Input("num_lower: int64").Input("num_upper: int64").Output("band: T").Attr("T: " NUMTYPES);
REGISTER_OP("BatchMatrixBandPart").Input("input: T").Input("num_lower: int64").Input("num_upper: int64").Output("band: T").Attr("T: " NUMTYPES);
REGISTER_OP("SpaceToDepth").Input("input: T").Output("output: T").Attr("T: type").Attr("block_size: int >= 2");
REGISTER_OP("DepthToSpace").Input("input: T").Output("output: T").Attr("T: type").Attr("block_size: int >= 2");
REGISTER_OP("MirrorPad").Input("input: T").Input("paddings: Tpaddings").Output("output: T").Attr("T: type").Attr("Tpaddings: {int32, int64} = int32").Attr("mode: string");
REGISTER_OP("ReverseSequence").Input("input: T").Input("seq_lengths: Tlen").Output("output: T").Attr("seq_dim: int").Attr("batch_dim: int = 0").Attr("T: " NUMTYPES).Attr("Tlen: {int32, int64} = int64");
REGISTER_OP("Bitcast").Input("input: T").Output("output: type").Attr("T: " NUMTYPES).Attr("type: " NUMTYPES);
// math breadth (trig / special functions / mod — reference math_ops.cc)
#define UOP(NAME) REGISTER_OP(NAME).Input("x: T").Output("y: T").Attr("T: " FLOATTYPES)
UOP("Tan");
UOP("Asin");
UOP("Acos");
UOP("Atan");
UOP("Erf");
UOP("Erfc");
UOP("Expm1");
UOP("Lgamma");
UOP("Digamma");
UOP("Rint");
UOP("Softsign");
UOP("Inv");
#undef UOP
REGISTER_OP("SoftsignGrad").Input("gradients: T").Input("features: T").Output("backprops: T").Attr("T: " FLOATTYPES);
REGISTER_OP("InvGrad").Input("y: T").Input("dy: T").Output("z: T").Attr("T: " FLOATTYPES);
REGISTER_OP("Mod").Input("x: T").Input("y: T").Output("z: T").Attr("T: {float, double, int32, int64}");
REGISTER_OP("ApproximateEqual").Input("x: T").Input("y: T").Output("z: bool").Attr("T: " NUMTYPES).Attr("tolerance: float = 1e-05");
REGISTER_OP("AsString").Input("input: T").Output("output: string").Attr("T: {float, double, int32, int64, bool}").Attr("precision: int = -1").Attr("scientific: bool = false").Attr("shortest: bool = false").Attr("width: int = -1").Attr("fill: string = ''");
REGISTER_OP("DecodeRaw").Input("bytes: string").Output("output: out_type").Attr("out_type: {float, double, int32, uint8, int16, int8, int64}").Attr("little_endian: bool = true");
// string family (reference core/ops/string_ops.cc; kernels/cpu_strings.cc)
REGISTER_OP("StringJoin").Input("inputs: N * string").Output("output: string").Attr("N: int >= 1").Attr("separator: string = ''");
REGISTER_OP("StringSplit").Input("input: string").Input("delimiter: string").Output("indices: int64").Output("values: string").Output("shape: int64").Attr("skip_empty: bool = true");
REGISTER_OP("Substr").Input("input: string").Input("pos: T").Input("len: T").Output("output: string").Attr("T: {int32, int64}");
REGISTER_OP("StringToHashBucket").Input("string_tensor: string").Output("output: int64").Attr("num_buckets: int >= 1");
REGISTER_OP("StringToHashBucketFast").Input("input: string").Output("output: int64").Attr("num_buckets: int >= 1");
REGISTER_OP("StringToHashBucketStrong").Input("input: string").Output("output: int64").Attr("num_buckets: int >= 1").Attr("key: list(int)");
REGISTER_OP("StringToNumber").Input("string_tensor: string").Output("output: out_type").Attr("out_type: {float, double, int32, int64} = float");
REGISTER_OP("ReduceJoin").Input("inputs: string").Input("reduction_indices: int32").Output("output: string").Attr("keep_dims: bool = false").Attr("separator: string = ''");
REGISTER_OP("EncodeBase64").Input("input: string").Output("output: string").Attr("pad: bool = false");
REGISTER_OP("DecodeBase64").Input("input: string").Output("output: string");
// lookup tables (reference core/ops/data_flow_ops.cc lookup section)
REGISTER_OP("HashTable").Output("table_handle: Ref(string)").Attr("container: string = ''").Attr("shared_name: string = ''").Attr("key_dtype: type").Attr("value_dtype: type").SetIsStateful();
REGISTER_OP("MutableHashTable").Output("table_handle: Ref(string)").Attr("container: string = ''").Attr("shared_name: string = ''").Attr("key_dtype: type").Attr("value_dtype: type").SetIsStateful();
REGISTER_OP("InitializeTable").Input("table_handle: Ref(string)").Input("keys: Tkey").Input("values: Tval").Attr("Tkey: type").Attr("Tval: type").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("LookupTableInsert").Input("table_handle: Ref(string)").Input("keys: Tin").Input("values: Tout").Attr("Tin: type").Attr("Tout: type").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("LookupTableImport").Input("table_handle: Ref(string)").Input("keys: Tin").Input("values: Tout").Attr("Tin: type").Attr("Tout: type").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("LookupTableFind").Input("table_handle: Ref(string)").Input("keys: Tin").Input("default_value: Tout").Output("values: Tout").Attr("Tin: type").Attr("Tout: type").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("LookupTableSize").Input("table_handle: Ref(string)").Output("size: int64").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("LookupTableExport").Input("table_handle: Ref(string)").Output("keys: Tkeys").Output("values: Tvalues").Attr("Tkeys: type").Attr("Tvalues: type").SetIsStateful().SetAllowsUninitializedInput();
// stacks (reference stack_ops.cc)
REGISTER_OP("Stack").Output("handle: Ref(string)").Attr("elem_type: type").Attr("stack_name: string = ''").SetIsStateful();
REGISTER_OP("StackPush").Input("handle: Ref(string)").Input("elem: T").Output("output: T").Attr("T: type").Attr("swap_memory: bool = false").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("StackPop").Input("handle: Ref(string)").Output("elem: elem_type").Attr("elem_type: type").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("StackClose").Input("handle: Ref(string)").SetIsStateful().SetAllowsUninitializedInput();
// barrier (reference barrier_ops.cc)
REGISTER_OP("Barrier").Output("handle: Ref(string)").Attr("component_types: list(type) >= 1").Attr("shapes: list(shape) = []").Attr("capacity: int = -1").Attr("container: string = ''").Attr("shared_name: string = ''").SetIsStateful();
REGISTER_OP("BarrierInsertMany").Input("handle: Ref(string)").Input("keys: string").Input("values: T").Attr("T: type").Attr("component_index: int").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("BarrierTakeMany").Input("handle: Ref(string)").Input("num_elements: int32").Output("indices: int64").Output("keys: string").Output("values: component_types").Attr("component_types: list(type) >= 1").Attr("allow_small_batch: bool = false").Attr("wait_for_incomplete: bool = false").Attr("timeout_ms: int = -1").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("BarrierClose").Input("handle: Ref(string)").Attr("cancel_pending_enqueues: bool = false").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("BarrierReadySize").Input("handle: Ref(string)").Output("size: int32").SetIsStateful().SetAllowsUninitializedInput();
REGISTER_OP("BarrierIncompleteSize").Input("handle: Ref(string)").Output("size: int32").SetIsStateful().SetAllowsUninitializedInput();
// variable scatter updates
#define SCATTER_VAR_OP(NAME) REGISTER_OP(NAME).Input("ref: Ref(T)").Input("indices: Tindices").Input("updates: T").Output("output_ref: Ref(T)").Attr("T: " NUMTYPES).Attr("Tindices: {int32, int64}").Attr("use_locking: bool = true")
SCATTER_VAR_OP("ScatterUpdate");
SCATTER_VAR_OP("ScatterMul");
SCATTER_VAR_OP("ScatterDiv");
#undef SCATTER_VAR_OP

}  // namespace stf
