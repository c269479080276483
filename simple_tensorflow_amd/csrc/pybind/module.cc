// pybind11 bridge: the analog of the reference's SWIG pywrap_tensorflow +
// tf_session_helper.cc (numpy <-> Tensor, session lifecycle, op registry
// introspection for the Python op_def_library).
#include <pybind11/numpy.h>
#include <pybind11/complex.h>

#include <algorithm>
#include <complex>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <map>
#include <mutex>

#include <fstream>

#include "core/pb.h"
#include "core/protos.h"
#include "util/table.h"
#include "framework/op.h"
#include "runtime/session.h"

namespace py = pybind11;
using namespace stf;

namespace {

DataType NumpyToDataType(const py::array& a) {
  auto dt = a.dtype();
  char kind = dt.kind();
  int size = (int)dt.itemsize();
  if (kind == 'f' && size == 4) return DT_FLOAT;
  if (kind == 'f' && size == 8) return DT_DOUBLE;
  if (kind == 'f' && size == 2) return DT_HALF;
  if (kind == 'i' && size == 4) return DT_INT32;
  if (kind == 'i' && size == 8) return DT_INT64;
  if (kind == 'i' && size == 1) return DT_INT8;
  if (kind == 'i' && size == 2) return DT_INT16;
  if (kind == 'u' && size == 1) return DT_UINT8;
  if (kind == 'u' && size == 2) return DT_UINT16;  // also bf16 carrier
  if (kind == 'b') return DT_BOOL;
  if (kind == 'c' && size == 8) return DT_COMPLEX64;
  if (kind == 'c' && size == 16) return DT_COMPLEX128;
  return DT_INVALID;
}

Tensor NumpyToTensor(py::array arr) {
  DataType dt = NumpyToDataType(arr);
  if (dt == DT_INVALID)
    throw std::runtime_error("Unsupported numpy dtype for feed");
  auto buf = py::array::ensure(arr, py::array::c_style | py::array::forcecast);
  TensorShape shape;
  for (int i = 0; i < buf.ndim(); ++i) shape.AddDim(buf.shape(i));
  Tensor t(dt, shape);
  std::memcpy(t.raw_data(), buf.data(), t.TotalBytes());
  return t;
}

py::object TensorToPy(const Tensor& t) {
  std::vector<ssize_t> shape;
  for (auto d : t.shape().dim_sizes()) shape.push_back((ssize_t)d);
  std::string fmt;
  switch (t.dtype()) {
    case DT_FLOAT: fmt = py::format_descriptor<float>::format(); break;
    case DT_DOUBLE: fmt = py::format_descriptor<double>::format(); break;
    case DT_INT32: fmt = py::format_descriptor<int32_t>::format(); break;
    case DT_INT64: fmt = py::format_descriptor<int64_t>::format(); break;
    case DT_UINT8: fmt = py::format_descriptor<uint8_t>::format(); break;
    case DT_INT8: fmt = py::format_descriptor<int8_t>::format(); break;
    case DT_INT16: fmt = py::format_descriptor<int16_t>::format(); break;
    case DT_UINT16:
    case DT_BFLOAT16:
    case DT_HALF: fmt = py::format_descriptor<uint16_t>::format(); break;
    case DT_BOOL: fmt = py::format_descriptor<bool>::format(); break;
    case DT_COMPLEX64: {
      py::array outc(py::dtype::of<std::complex<float>>(), shape);
      std::memcpy(outc.mutable_data(), t.raw_data(),
                  (size_t)t.NumElements() * 8);
      return outc;
    }
    case DT_COMPLEX128: {
      py::array outc(py::dtype::of<std::complex<double>>(), shape);
      std::memcpy(outc.mutable_data(), t.raw_data(),
                  (size_t)t.NumElements() * 16);
      return outc;
    }
    case DT_STRING: {
      // Return list (or scalar) of bytes.
      const std::string* p = t.flat<std::string>();
      if (t.dims() == 0) return py::bytes(p[0]);
      py::list out;
      for (int64_t i = 0; i < t.NumElements(); ++i)
        out.append(py::bytes(p[i]));
      return out;
    }
    default:
      throw std::runtime_error(std::string("Unsupported fetch dtype ") +
                               DataTypeString(t.dtype()));
  }
  size_t es = DataTypeSize(t.dtype());
  py::array out(py::dtype(fmt), shape);
  std::memcpy(out.mutable_data(), t.raw_data(), (size_t)t.NumElements() * es);
  if (t.dtype() == DT_BFLOAT16) {
    // convert bf16 -> float32 for numpy friendliness
    py::array_t<float> f(shape);
    const uint16_t* src = t.flat<uint16_t>();
    float* dst = (float*)f.mutable_data();
    for (int64_t i = 0; i < t.NumElements(); ++i) {
      uint32_t bits = ((uint32_t)src[i]) << 16;
      std::memcpy(dst + i, &bits, 4);
    }
    return f;
  }
  return out;
}

py::dict AttrValueToPy(const AttrValue& v);

py::dict OpDefToPy(const OpDef& op) {
  py::dict d;
  d["name"] = op.name;
  auto args = [](const std::vector<OpDef::ArgDef>& as) {
    py::list out;
    for (auto& a : as) {
      py::dict ad;
      ad["name"] = a.name;
      ad["type"] = (int)a.type;
      ad["type_attr"] = a.type_attr;
      ad["number_attr"] = a.number_attr;
      ad["type_list_attr"] = a.type_list_attr;
      ad["is_ref"] = a.is_ref;
      out.append(ad);
    }
    return out;
  };
  d["input_arg"] = args(op.input_arg);
  d["output_arg"] = args(op.output_arg);
  py::list attrs;
  for (auto& a : op.attr) {
    py::dict ad;
    ad["name"] = a.name;
    ad["type"] = a.type;
    ad["has_default"] = a.has_default;
    if (a.has_default) ad["default"] = AttrValueToPy(a.default_value);
    py::list allowed;
    for (auto t : a.allowed) allowed.append((int)t);
    ad["allowed"] = allowed;
    attrs.append(ad);
  }
  d["attr"] = attrs;
  d["is_stateful"] = op.is_stateful;
  return d;
}

py::dict AttrValueToPy(const AttrValue& v) {
  py::dict d;
  d["kind"] = std::string(1, v.kind ? v.kind : '0');
  switch (v.kind) {
    case 's': d["value"] = py::bytes(v.s); break;
    case 'i': d["value"] = v.i; break;
    case 'f': d["value"] = v.f; break;
    case 'b': d["value"] = v.b; break;
    case 't': d["value"] = (int)v.type; break;
    case 'l': {
      py::dict lv;
      lv["s"] = v.list.s;
      lv["i"] = v.list.i;
      lv["f"] = v.list.f;
      lv["b"] = v.list.b;
      lv["type"] = v.list.type;
      d["value"] = lv;
      break;
    }
    default: d["value"] = py::none();
  }
  return d;
}

// ---- py_func bridge (reference python/lib/core/py_func.cc + kernels
// script_ops): python callables registered by token, invoked from the
// executor with the GIL re-acquired. ----
std::mutex g_pyfunc_mu;
std::map<std::string, py::function>& PyFuncRegistry() {
  static auto* m = new std::map<std::string, py::function>();
  return *m;
}

class PyFuncOp : public OpKernel {
 public:
  explicit PyFuncOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("token", &token_);
    auto it = c->def().attr.find("Tout");
    if (it != c->def().attr.end())
      for (int t : it->second.list.type) tout_.push_back((DataType)t);
  }
  void Compute(OpKernelContext* ctx) override {
    py::gil_scoped_acquire gil;
    py::function fn;
    {
      std::lock_guard<std::mutex> l(g_pyfunc_mu);
      auto it = PyFuncRegistry().find(token_);
      if (it == PyFuncRegistry().end()) {
        ctx->SetStatus(errors::NotFound("py_func token ", token_));
        return;
      }
      fn = it->second;
    }
    py::list args;
    for (int i = 0; i < ctx->num_inputs(); ++i)
      args.append(TensorToPy(ctx->input(i)));
    py::object result;
    try {
      result = fn(*py::tuple(args));
    } catch (py::error_already_set& e) {
      ctx->SetStatus(errors::Internal("py_func raised: ", e.what()));
      return;
    }
    py::list outs;
    if (py::isinstance<py::tuple>(result) || py::isinstance<py::list>(result))
      outs = py::list(result);
    else if (!result.is_none())
      outs.append(result);
    if ((size_t)outs.size() != tout_.size()) {
      ctx->SetStatus(errors::InvalidArgument(
          "py_func returned ", (int)outs.size(), " values, expected ",
          (int)tout_.size()));
      return;
    }
    for (size_t i = 0; i < tout_.size(); ++i) {
      py::object o = outs[i];
      if (tout_[i] == DT_STRING) {
        Tensor t(DT_STRING, TensorShape({}));
        t.flat<std::string>()[0] = py::cast<std::string>(o);
        ctx->set_output((int)i, t);
        continue;
      }
      py::array arr = py::array::ensure(o);
      Tensor t = NumpyToTensor(arr);
      if (t.dtype() != tout_[i]) {
        ctx->SetStatus(errors::InvalidArgument(
            "py_func output ", (int)i, " dtype mismatch"));
        return;
      }
      ctx->set_output((int)i, t);
    }
  }

 private:
  std::string token_;
  std::vector<DataType> tout_;
};
REGISTER_KERNEL_BUILDER(Name("PyFunc").Device(DEVICE_CPU), PyFuncOp);
REGISTER_KERNEL_BUILDER(Name("PyFuncStateless").Device(DEVICE_CPU), PyFuncOp);

// CPU fallback for the RCCL collective ops: delegates to a python callable
// (registered by parallel/dist.py, backed by torch.distributed gloo) so the
// multi-process data-parallel graph runs — and is tested — on CPU-only
// hosts. On GPU boxes the HIP RCCL kernels in rccl/rccl_ops.cc are used.
class CpuCollectiveOp : public OpKernel {
 public:
  CpuCollectiveOp(OpKernelConstruction* c, const char* token)
      : OpKernel(c), token_(token) {}
  void Compute(OpKernelContext* ctx) override {
    py::gil_scoped_acquire gil;
    py::function fn;
    {
      std::lock_guard<std::mutex> l(g_pyfunc_mu);
      auto it = PyFuncRegistry().find(token_);
      if (it == PyFuncRegistry().end()) {
        ctx->SetStatus(errors::FailedPrecondition(
            "CPU collective fallback not initialized (", token_,
            ") — call parallel.dist.init() first"));
        return;
      }
      fn = it->second;
    }
    py::object result;
    try {
      result = fn(TensorToPy(ctx->input(0)));
    } catch (py::error_already_set& e) {
      ctx->SetStatus(errors::Internal("collective raised: ", e.what()));
      return;
    }
    py::array arr = py::array::ensure(result);
    Tensor t = NumpyToTensor(arr);
    if (t.dtype() != ctx->input(0).dtype()) {
      // gloo reduces in f32; cast back to the graph dtype (e.g. bf16)
      Tensor out(ctx->input(0).dtype(), t.shape());
      if (ctx->input(0).dtype() == DT_BFLOAT16 && t.dtype() == DT_FLOAT) {
        const float* src = t.flat<float>();
        bfloat16* dst = out.flat<bfloat16>();
        for (int64_t i = 0; i < t.NumElements(); ++i)
          dst[i] = bfloat16(src[i]);
        ctx->set_output(0, out);
        return;
      }
      ctx->SetStatus(errors::Internal("collective dtype mismatch"));
      return;
    }
    ctx->set_output(0, t);
  }

 private:
  std::string token_;
};
class CpuAllReduceOp : public CpuCollectiveOp {
 public:
  explicit CpuAllReduceOp(OpKernelConstruction* c)
      : CpuCollectiveOp(c, "__cpu_collective_allreduce") {}
};
class CpuBroadcastOp : public CpuCollectiveOp {
 public:
  explicit CpuBroadcastOp(OpKernelConstruction* c)
      : CpuCollectiveOp(c, "__cpu_collective_broadcast") {}
};
REGISTER_KERNEL_BUILDER(Name("RcclAllReduce").Device(DEVICE_CPU),
                        CpuAllReduceOp);
REGISTER_KERNEL_BUILDER(Name("RcclBroadcast").Device(DEVICE_CPU),
                        CpuBroadcastOp);

// CPU fallback of the fused-bucket path: pack all segments into one f32
// array, ONE collective call per bucket (same wire semantics as the GPU
// ncclAllReduce of the flat staging buffer), unpack with the 1/world scale.
class CpuBucketAllReduceOp : public OpKernel {
 public:
  explicit CpuBucketAllReduceOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("scale", &scale_);
  }
  void Compute(OpKernelContext* ctx) override {
    int n = ctx->num_inputs();
    int64_t total = 0;
    for (int i = 0; i < n; ++i) total += ctx->input(i).NumElements();
    std::vector<float> flat((size_t)total);
    int64_t off = 0;
    for (int i = 0; i < n; ++i) {
      const Tensor& in = ctx->input(i);
      int64_t m = in.NumElements();
      if (in.dtype() == DT_FLOAT) {
        memcpy(flat.data() + off, in.raw_data(), m * sizeof(float));
      } else if (in.dtype() == DT_BFLOAT16) {
        const bfloat16* src = in.flat<bfloat16>();
        for (int64_t j = 0; j < m; ++j)
          flat[off + j] = (float)src[j];
      } else {
        ctx->SetStatus(errors::InvalidArgument("bucket dtype must be f32/bf16"));
        return;
      }
      off += m;
    }
    py::gil_scoped_acquire gil;
    py::function fn;
    {
      std::lock_guard<std::mutex> l(g_pyfunc_mu);
      auto it = PyFuncRegistry().find("__cpu_collective_allreduce");
      if (it == PyFuncRegistry().end()) {
        ctx->SetStatus(errors::FailedPrecondition(
            "CPU collective fallback not initialized — call "
            "parallel.dist.init() first"));
        return;
      }
      fn = it->second;
    }
    py::array_t<float> arr((py::ssize_t)total);
    memcpy(arr.mutable_data(), flat.data(), total * sizeof(float));
    py::object result;
    try {
      result = fn(arr);
    } catch (py::error_already_set& e) {
      ctx->SetStatus(errors::Internal("collective raised: ", e.what()));
      return;
    }
    py::array red = py::array::ensure(result);
    Tensor rt = NumpyToTensor(red);
    if (rt.dtype() != DT_FLOAT || rt.NumElements() != total) {
      ctx->SetStatus(errors::Internal("bucket collective shape mismatch"));
      return;
    }
    const float* rf = rt.flat<float>();
    off = 0;
    for (int i = 0; i < n; ++i) {
      const Tensor& in = ctx->input(i);
      int64_t m = in.NumElements();
      Tensor* out = ctx->allocate_output(i, in.shape());
      if (in.dtype() == DT_FLOAT) {
        float* dst = out->flat<float>();
        for (int64_t j = 0; j < m; ++j) dst[j] = rf[off + j] * scale_;
      } else {
        bfloat16* dst = out->flat<bfloat16>();
        for (int64_t j = 0; j < m; ++j)
          dst[j] = bfloat16(rf[off + j] * scale_);
      }
      off += m;
    }
  }

 private:
  float scale_ = 1.0f;
};
REGISTER_KERNEL_BUILDER(Name("RcclBucketAllReduce").Device(DEVICE_CPU),
                        CpuBucketAllReduceOp);

// Everything on CPU is synchronous already; the sync is a no-op.
class CpuCommSyncOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {}
};
REGISTER_KERNEL_BUILDER(Name("RcclCommSync").Device(DEVICE_CPU),
                        CpuCommSyncOp);

class PySession;
namespace {
std::mutex g_sessions_mu;
std::vector<PySession*> g_sessions;  // live sessions, for Session.reset
}  // namespace

class PySession {
 public:
  explicit PySession(bool cpu_only) : sess_(cpu_only) {
    std::lock_guard<std::mutex> l(g_sessions_mu);
    g_sessions.push_back(this);
  }
  ~PySession() {
    std::lock_guard<std::mutex> l(g_sessions_mu);
    g_sessions.erase(std::remove(g_sessions.begin(), g_sessions.end(), this),
                     g_sessions.end());
  }

  void Create(py::bytes graph_def) {
    GraphDef gd;
    std::string data = graph_def;
    if (!gd.ParseFromString(data))
      throw std::runtime_error("Failed to parse GraphDef");
    Status s = sess_.Create(gd);
    if (!s.ok()) throw std::runtime_error(s.ToString());
  }
  void Extend(py::bytes graph_def) {
    GraphDef gd;
    std::string data = graph_def;
    if (!gd.ParseFromString(data))
      throw std::runtime_error("Failed to parse GraphDef");
    Status s = sess_.Extend(gd);
    if (!s.ok()) throw std::runtime_error(s.ToString());
  }

  // Per-node host-side timings from the last Run(collect_stats=true):
  // list of (node, op, start_us, end_us).
  py::list LastStats() {
    py::list out;
    for (auto& ns : last_stats_)
      out.append(py::make_tuple(ns.node, ns.op, ns.start_us, ns.end_us,
                                ns.device));
    return out;
  }

  void Reset() { sess_.Reset(); }

  std::string PartialRunSetup(std::vector<std::string> feeds,
                              std::vector<std::string> fetches,
                              std::vector<std::string> targets) {
    std::string handle;
    Status s = sess_.PartialRunSetup(feeds, fetches, targets, &handle);
    if (!s.ok()) throw std::runtime_error(s.ToString());
    return handle;
  }

  py::list PartialRun(const std::string& handle, py::dict feeds,
                      std::vector<std::string> fetches) {
    std::vector<std::pair<std::string, Tensor>> feed_vec;
    for (auto item : feeds) {
      std::string name = py::cast<std::string>(item.first);
      py::object val = py::reinterpret_borrow<py::object>(item.second);
      feed_vec.emplace_back(name, NumpyToTensor(py::cast<py::array>(val)));
    }
    std::vector<Tensor> outputs;
    Status s;
    {
      py::gil_scoped_release release;
      s = sess_.PartialRun(handle, feed_vec, fetches, &outputs);
    }
    if (!s.ok()) throw std::runtime_error(s.ToString());
    py::list out;
    for (auto& t : outputs) out.append(TensorToPy(t));
    return out;
  }

  py::list Run(py::dict feeds, std::vector<std::string> fetches,
               std::vector<std::string> targets, bool collect_stats = false,
               int64_t timeout_ms = 0) {
    std::vector<std::pair<std::string, Tensor>> feed_vec;
    for (auto item : feeds) {
      std::string name = py::cast<std::string>(item.first);
      py::object val = py::reinterpret_borrow<py::object>(item.second);
      if (py::isinstance<py::bytes>(val) || py::isinstance<py::str>(val)) {
        Tensor t(DT_STRING, TensorShape({}));
        t.flat<std::string>()[0] = py::cast<std::string>(val);
        feed_vec.emplace_back(name, t);
      } else {
        feed_vec.emplace_back(name, NumpyToTensor(py::cast<py::array>(val)));
      }
    }
    std::vector<Tensor> outputs;
    Status s;
    StatsCollector stats;
    {
      py::gil_scoped_release release;
      s = sess_.Run(feed_vec, fetches, targets, &outputs,
                    collect_stats ? &stats : nullptr, timeout_ms);
    }
    if (!s.ok()) throw std::runtime_error(s.ToString());
    if (collect_stats) last_stats_ = std::move(stats.stats);
    py::list out;
    for (auto& t : outputs) out.append(TensorToPy(t));
    return out;
  }

  void Sync() {
    py::gil_scoped_release release;
    Status s = sess_.SyncAllDevices();
    if (!s.ok()) throw std::runtime_error(s.ToString());
  }

  int NumGpus() {
    int n = 0;
    for (auto& d : sess_.device_mgr()->devices())
      if (d->is_gpu()) ++n;
    return n;
  }

 private:
  DirectSession sess_;
  std::vector<NodeStats> last_stats_;
};

}  // namespace

namespace stf {
std::string HipGraphSelfTest();
std::string RcclGetUniqueId();
Status RcclInit(int nranks, int rank, const std::string& id_bytes);
}

PYBIND11_MODULE(_core, m) {
  m.doc() = "simple_tensorflow_amd core runtime (MI355X-native)";
  // checkpoint introspection (reference python/tools/inspect_checkpoint.py
  // via checkpoint_reader.cc): list (name, dtype_enum, shape) entries of a
  // V2 bundle index.
  m.def("list_checkpoint", [](const std::string& prefix) {
    std::ifstream f(prefix + ".index", std::ios::binary);
    if (!f) throw std::runtime_error("checkpoint index not found: " + prefix);
    std::string data((std::istreambuf_iterator<char>(f)),
                     std::istreambuf_iterator<char>());
    std::map<std::string, std::string> index;
    Status s = table::ReadTable(data, &index);
    if (!s.ok()) throw std::runtime_error(s.ToString());
    py::list out;
    for (auto& kv : index) {
      if (kv.first.empty()) continue;  // header entry
      pb::Reader r(kv.second);
      int field, wire;
      int dtype = 0;
      std::vector<int64_t> shape;
      bool ok_entry = true;
      while (r.ReadTag(&field, &wire)) {
        if (field == 1) {
          uint64_t v;
          if (!r.ReadVarint(&v)) { ok_entry = false; break; }
          dtype = (int)v;
        } else if (field == 2) {
          const char* d;
          size_t l;
          if (!r.ReadView(&d, &l)) { ok_entry = false; break; }
          pb::Reader sub(d, l);
          TensorShapeProto p;
          if (p.Parse(&sub)) {
            TensorShape ts = p.AsShape();
            for (auto dim : ts.dim_sizes()) shape.push_back(dim);
          }
        } else if (!r.SkipField(wire)) {
          ok_entry = false;
          break;
        }
      }
      if (ok_entry && dtype != 0)
        out.append(py::make_tuple(kv.first, dtype, shape));
    }
    return out;
  });
  m.def("register_py_func", [](const std::string& token, py::function fn) {
    std::lock_guard<std::mutex> l(g_pyfunc_mu);
    PyFuncRegistry()[token] = std::move(fn);
  });
  m.def("remove_py_func", [](const std::string& token) {
    std::lock_guard<std::mutex> l(g_pyfunc_mu);
    PyFuncRegistry().erase(token);
  });

  py::class_<PySession>(m, "Session")
      .def(py::init<bool>(), py::arg("cpu_only") = false)
      .def("create", &PySession::Create)
      .def("extend", &PySession::Extend)
      .def("run", &PySession::Run, py::arg("feeds"), py::arg("fetches"),
           py::arg("targets"), py::arg("collect_stats") = false,
           py::arg("timeout_ms") = 0)
      .def("reset", &PySession::Reset)
      .def("partial_run_setup", &PySession::PartialRunSetup)
      .def("partial_run", &PySession::PartialRun)
      .def_static("reset_all", []() {
        // Session.reset analog for in-process sessions: reset the stateful
        // containers of EVERY live session (reference TF_Reset clears the
        // target's session registry).
        std::lock_guard<std::mutex> l(g_sessions_mu);
        for (PySession* s : g_sessions) s->Reset();
      })
      .def("last_stats", &PySession::LastStats)
      .def("sync", &PySession::Sync)
      .def("num_gpus", &PySession::NumGpus);

  m.def("list_ops", []() {
    py::dict out;
    for (auto& name : OpRegistry::Global()->ListOps()) {
      out[py::str(name)] = OpDefToPy(*OpRegistry::Global()->LookUp(name));
    }
    return out;
  });
  m.def("hipgraph_selftest", []() -> std::string {
    // Probe: capture with cross-thread enqueue + an in-capture hipMalloc —
    // mirrors what the executor does during a captured step.
    return stf::HipGraphSelfTest();
  });
  m.def("rccl_get_unique_id", []() {
    return py::bytes(stf::RcclGetUniqueId());
  });
  m.def("rccl_init", [](int nranks, int rank, py::bytes id) {
    Status s = stf::RcclInit(nranks, rank, std::string(id));
    if (!s.ok()) throw std::runtime_error(s.ToString());
  });
  m.def("has_gpu", []() {
    // cheap probe without creating a session
    return false;
  });
}
