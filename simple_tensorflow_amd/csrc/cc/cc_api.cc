// Scope / op builders / ClientSession (see cc_api.h; reference
// cc/framework/scope.cc + cc/client/client_session.cc).
#include "cc/cc_api.h"

#include "framework/op.h"
#include "runtime/session.h"

namespace stf {
namespace cc {

Scope Scope::NewRootScope() {
  return Scope(std::make_shared<GraphState>(), "");
}

Scope Scope::NewSubScope(const std::string& child) const {
  std::string p = prefix_.empty() ? child : prefix_ + "/" + child;
  return Scope(state_, p);
}

std::string Scope::UniqueName(const std::string& base) const {
  std::string full = prefix_.empty() ? base : prefix_ + "/" + base;
  int& n = state_->names_used[full];
  std::string name = n == 0 ? full : full + "_" + std::to_string(n);
  n++;
  return name;
}

Output Scope::AddOp(const std::string& op_type, const std::string& name,
                    const std::vector<Output>& inputs,
                    const std::map<std::string, AttrValue>& attrs) const {
  if (!state_->status.ok()) return Output();
  const OpDef* od = OpRegistry::Global()->LookUp(op_type);
  if (!od) {
    state_->status = errors::NotFound("Op not registered: ", op_type);
    return Output();
  }
  NodeDef nd;
  nd.name = UniqueName(name.empty() ? op_type : name);
  nd.op = op_type;
  for (auto& in : inputs) nd.input.push_back(in.name());
  nd.attr = attrs;
  // Fill defaulted attrs the kernels require (e.g. T) from context is the
  // caller's job; defaults with has_default are resolved at kernel build.
  state_->gdef.node.push_back(nd);
  return Output(nd.name, 0);
}

namespace {
const NodeDef* FindNode(const Scope& s, const std::string& name) {
  for (auto& n : s.state()->gdef.node)
    if (n.name == name) return &n;
  return nullptr;
}

DataType OutDType(const Scope& s, const Output& o) {
  const NodeDef* n = FindNode(s, o.node);
  if (!n) return DT_FLOAT;
  auto it = n->attr.find("T");
  if (it != n->attr.end() && it->second.kind == 't') return it->second.type;
  it = n->attr.find("dtype");
  if (it != n->attr.end() && it->second.kind == 't') return it->second.type;
  return DT_FLOAT;
}
}  // namespace

namespace ops {

Output Const(const Scope& s, const Tensor& value, const std::string& name) {
  AttrValue v;
  v.kind = 'e';
  value.AsProto(&v.tensor);
  AttrValue t = AttrValue::Type(value.dtype());
  return s.AddOp("Const", name, {}, {{"dtype", t}, {"value", v}});
}

Output Const(const Scope& s, float value) {
  Tensor t(DT_FLOAT, TensorShape({}));
  t.flat<float>()[0] = value;
  return Const(s, t);
}

Output Const(const Scope& s, const std::vector<float>& value,
             const std::vector<int64_t>& shape) {
  TensorShape sh;
  for (auto d : shape) sh.AddDim(d);
  Tensor t(DT_FLOAT, sh);
  for (size_t i = 0; i < value.size(); ++i) t.flat<float>()[i] = value[i];
  return Const(s, t);
}

Output Placeholder(const Scope& s, DataType dtype, const std::string& name) {
  return s.AddOp("Placeholder", name, {},
                 {{"dtype", AttrValue::Type(dtype)}});
}

static Output Binary(const Scope& s, const char* op, Output a, Output b) {
  return s.AddOp(op, op, {a, b}, {{"T", AttrValue::Type(OutDType(s, a))}});
}
static Output Unary(const Scope& s, const char* op, Output a) {
  return s.AddOp(op, op, {a}, {{"T", AttrValue::Type(OutDType(s, a))}});
}

Output Add(const Scope& s, Output a, Output b) { return Binary(s, "Add", a, b); }
Output Sub(const Scope& s, Output a, Output b) { return Binary(s, "Sub", a, b); }
Output Mul(const Scope& s, Output a, Output b) { return Binary(s, "Mul", a, b); }
Output Div(const Scope& s, Output a, Output b) { return Binary(s, "Div", a, b); }
Output Neg(const Scope& s, Output a) { return Unary(s, "Neg", a); }
Output Square(const Scope& s, Output a) { return Unary(s, "Square", a); }
Output Relu(const Scope& s, Output a) { return Unary(s, "Relu", a); }
Output Tanh(const Scope& s, Output a) { return Unary(s, "Tanh", a); }
Output Sigmoid(const Scope& s, Output a) { return Unary(s, "Sigmoid", a); }
Output Exp(const Scope& s, Output a) { return Unary(s, "Exp", a); }
Output Identity(const Scope& s, Output a) { return Unary(s, "Identity", a); }
Output OnesLike(const Scope& s, Output a) { return Unary(s, "OnesLike", a); }
Output ZerosLike(const Scope& s, Output a) { return Unary(s, "ZerosLike", a); }

Output MatMul(const Scope& s, Output a, Output b, bool transpose_a,
              bool transpose_b) {
  return s.AddOp("MatMul", "MatMul", {a, b},
                 {{"T", AttrValue::Type(OutDType(s, a))},
                  {"transpose_a", AttrValue::B(transpose_a)},
                  {"transpose_b", AttrValue::B(transpose_b)}});
}

Output Softmax(const Scope& s, Output logits) {
  return Unary(s, "Softmax", logits);
}

static Output AxesConst(const Scope& s, const std::vector<int>& axes) {
  Tensor t(DT_INT32, TensorShape({(int64_t)axes.size()}));
  for (size_t i = 0; i < axes.size(); ++i) t.flat<int32_t>()[i] = axes[i];
  return Const(s, t);
}

Output ReduceSum(const Scope& s, Output a, const std::vector<int>& axes) {
  return s.AddOp("Sum", "Sum", {a, AxesConst(s, axes)},
                 {{"T", AttrValue::Type(OutDType(s, a))},
                  {"keep_dims", AttrValue::B(false)}});
}

Output ReduceMean(const Scope& s, Output a, const std::vector<int>& axes) {
  return s.AddOp("Mean", "Mean", {a, AxesConst(s, axes)},
                 {{"T", AttrValue::Type(OutDType(s, a))},
                  {"keep_dims", AttrValue::B(false)}});
}

Output Reshape(const Scope& s, Output a, const std::vector<int64_t>& shape) {
  Tensor t(DT_INT32, TensorShape({(int64_t)shape.size()}));
  for (size_t i = 0; i < shape.size(); ++i)
    t.flat<int32_t>()[i] = (int32_t)shape[i];
  return s.AddOp("Reshape", "Reshape", {a, Const(s, t)},
                 {{"T", AttrValue::Type(OutDType(s, a))}});
}

}  // namespace ops

// ---------------------------------------------------------------------------
// ClientSession
// ---------------------------------------------------------------------------
ClientSession::ClientSession(const Scope& scope)
    : scope_(scope), sess_(new DirectSession()) {}
ClientSession::~ClientSession() {}

Status ClientSession::Run(const std::vector<Output>& fetches,
                          std::vector<Tensor>* outputs) {
  return Run({}, fetches, {}, outputs);
}

Status ClientSession::Run(const std::vector<std::pair<Output, Tensor>>& feeds,
                          const std::vector<Output>& fetches,
                          std::vector<Tensor>* outputs) {
  return Run(feeds, fetches, {}, outputs);
}

Status ClientSession::Run(const std::vector<std::pair<Output, Tensor>>& feeds,
                          const std::vector<Output>& fetches,
                          const std::vector<Output>& targets,
                          std::vector<Tensor>* outputs) {
  STF_RETURN_IF_ERROR(scope_.status());
  GraphDef gd = scope_.ToGraphDef();
  if (gd.node.size() != created_nodes_) {
    STF_RETURN_IF_ERROR(sess_->Create(gd));
    created_nodes_ = gd.node.size();
  }
  std::vector<std::pair<std::string, Tensor>> f;
  for (auto& kv : feeds) f.emplace_back(kv.first.name(), kv.second);
  std::vector<std::string> fetch_names, target_names;
  for (auto& o : fetches) fetch_names.push_back(o.name());
  for (auto& o : targets) target_names.push_back(o.node);
  return sess_->Run(f, fetch_names, target_names, outputs);
}

}  // namespace cc
}  // namespace stf
