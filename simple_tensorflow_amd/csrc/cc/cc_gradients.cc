// C++ symbolic autodiff over a Scope's GraphDef (capability analog of
// reference cc/framework/gradients.cc AddSymbolicGradients +
// cc/framework/grad_op_registry.h + cc/gradients/{math,nn}_grad.cc).
// Reverse topological sweep with gradient accumulation via Add/AddN.
#include <algorithm>
#include <functional>
#include <set>

#include "cc/cc_api.h"

namespace stf {
namespace cc {
namespace {

struct NodeInfo {
  const NodeDef* def;
  std::vector<std::pair<std::string, int>> inputs;  // (node, port), data only
};

std::pair<std::string, int> ParseInput(const std::string& in) {
  auto colon = in.rfind(':');
  if (colon != std::string::npos &&
      in.find_first_not_of("0123456789", colon + 1) == std::string::npos &&
      colon + 1 < in.size())
    return {in.substr(0, colon), atoi(in.c_str() + colon + 1)};
  return {in, 0};
}

// Gradient function: given the node, its input Outputs and the gradient of
// its output 0, produce gradients for each data input ("" = no gradient).
using GradFn = std::function<Status(const Scope&, const NodeInfo&,
                                    const std::vector<Output>&,
                                    const Output&, std::vector<Output>*)>;

Status NoGrad(const Scope&, const NodeInfo& n, const std::vector<Output>& in,
              const Output&, std::vector<Output>* out) {
  out->assign(in.size(), Output());
  return Status::OK();
}

const std::map<std::string, GradFn>& GradRegistry() {
  static auto* reg = new std::map<std::string, GradFn>{
      {"Add",
       [](const Scope& s, const NodeInfo&, const std::vector<Output>& in,
          const Output& g, std::vector<Output>* out) {
         *out = {g, g};
         return Status::OK();
       }},
      {"Sub",
       [](const Scope& s, const NodeInfo&, const std::vector<Output>& in,
          const Output& g, std::vector<Output>* out) {
         *out = {g, ops::Neg(s, g)};
         return Status::OK();
       }},
      {"Mul",
       [](const Scope& s, const NodeInfo&, const std::vector<Output>& in,
          const Output& g, std::vector<Output>* out) {
         *out = {ops::Mul(s, g, in[1]), ops::Mul(s, g, in[0])};
         return Status::OK();
       }},
      {"Neg",
       [](const Scope& s, const NodeInfo&, const std::vector<Output>& in,
          const Output& g, std::vector<Output>* out) {
         *out = {ops::Neg(s, g)};
         return Status::OK();
       }},
      {"Identity",
       [](const Scope& s, const NodeInfo&, const std::vector<Output>& in,
          const Output& g, std::vector<Output>* out) {
         *out = {g};
         return Status::OK();
       }},
      {"Square",
       [](const Scope& s, const NodeInfo&, const std::vector<Output>& in,
          const Output& g, std::vector<Output>* out) {
         Output two = ops::Const(s, 2.0f);
         *out = {ops::Mul(s, g, ops::Mul(s, two, in[0]))};
         return Status::OK();
       }},
      {"Relu",
       [](const Scope& s, const NodeInfo& n, const std::vector<Output>& in,
          const Output& g, std::vector<Output>* out) {
         Output y(n.def->name, 0);
         *out = {s.AddOp("ReluGrad", "ReluGrad", {g, y},
                         {{"T", AttrValue::Type(DT_FLOAT)}})};
         return Status::OK();
       }},
      {"Tanh",
       [](const Scope& s, const NodeInfo& n, const std::vector<Output>& in,
          const Output& g, std::vector<Output>* out) {
         Output y(n.def->name, 0);
         Output one = ops::Const(s, 1.0f);
         *out = {ops::Mul(s, g, ops::Sub(s, one, ops::Square(s, y)))};
         return Status::OK();
       }},
      {"Exp",
       [](const Scope& s, const NodeInfo& n, const std::vector<Output>& in,
          const Output& g, std::vector<Output>* out) {
         *out = {ops::Mul(s, g, Output(n.def->name, 0))};
         return Status::OK();
       }},
      {"MatMul",
       [](const Scope& s, const NodeInfo& n, const std::vector<Output>& in,
          const Output& g, std::vector<Output>* out) {
         bool ta = false, tb = false;
         auto it = n.def->attr.find("transpose_a");
         if (it != n.def->attr.end()) ta = it->second.b;
         it = n.def->attr.find("transpose_b");
         if (it != n.def->attr.end()) tb = it->second.b;
         Output da, db;
         if (!ta && !tb) {
           da = ops::MatMul(s, g, in[1], false, true);
           db = ops::MatMul(s, in[0], g, true, false);
         } else if (!ta && tb) {
           da = ops::MatMul(s, g, in[1], false, false);
           db = ops::MatMul(s, g, in[0], true, false);
         } else if (ta && !tb) {
           da = ops::MatMul(s, in[1], g, false, true);
           db = ops::MatMul(s, in[0], g, false, false);
         } else {
           da = ops::MatMul(s, in[1], g, true, true);
           db = ops::MatMul(s, g, in[0], true, true);
         }
         *out = {da, db};
         return Status::OK();
       }},
      {"Sum",
       [](const Scope& s, const NodeInfo& n, const std::vector<Output>& in,
          const Output& g, std::vector<Output>* out) {
         // full reduce only: broadcast g back via Mul with OnesLike(x)
         *out = {ops::Mul(s, ops::OnesLike(s, in[0]), g), Output()};
         return Status::OK();
       }},
      {"Mean",
       [](const Scope& s, const NodeInfo& n, const std::vector<Output>& in,
          const Output& g, std::vector<Output>* out) {
         // g / N broadcast; N = Size(x) as float
         Output size = s.AddOp("Size", "Size", {in[0]},
                               {{"T", AttrValue::Type(DT_FLOAT)}});
         Output sizef = s.AddOp(
             "Cast", "Cast", {size},
             {{"SrcT", AttrValue::Type(DT_INT32)},
              {"DstT", AttrValue::Type(DT_FLOAT)}});
         Output scaled = ops::Div(s, g, sizef);
         *out = {ops::Mul(s, ops::OnesLike(s, in[0]), scaled), Output()};
         return Status::OK();
       }},
      {"Const", NoGrad},
      {"Placeholder", NoGrad},
  };
  return *reg;
}

}  // namespace

Status AddSymbolicGradients(const Scope& scope,
                            const std::vector<Output>& outputs,
                            const std::vector<Output>& inputs,
                            const std::vector<Output>& grad_inputs,
                            std::vector<Output>* grad_outputs) {
  // Snapshot the forward graph: gradient AddOps append to the same
  // GraphDef, which would invalidate iterators/pointers mid-sweep.
  std::vector<NodeDef> fwd = scope.ToGraphDef().node;
  std::map<std::string, NodeInfo> nodes;
  for (auto& n : fwd) {
    NodeInfo info;
    info.def = &n;
    for (auto& in : n.input) {
      if (!in.empty() && in[0] == '^') continue;
      info.inputs.push_back(ParseInput(in));
    }
    nodes[n.name] = info;
  }
  auto key_of = [](const std::string& node, int port) {
    return node + ":" + std::to_string(port);
  };
  // Reverse sweep from outputs. accum: "node:port" -> accumulated grad.
  std::map<std::string, Output> accum;
  for (size_t i = 0; i < outputs.size(); ++i) {
    Output g = i < grad_inputs.size() && grad_inputs[i].valid()
                   ? grad_inputs[i]
                   : ops::OnesLike(scope, outputs[i]);
    accum[key_of(outputs[i].node, outputs[i].index)] = g;
  }
  // GraphDef order is build order (topological for Scope-built graphs);
  // walk it backwards.
  for (auto it = fwd.rbegin(); it != fwd.rend(); ++it) {
    const NodeDef& n = *it;
    auto git = accum.find(key_of(n.name, 0));
    if (git == accum.end()) continue;
    Output gout = git->second;
    auto rit = GradRegistry().find(n.op);
    if (rit == GradRegistry().end())
      return errors::Unimplemented("No C++ gradient for op ", n.op);
    NodeInfo& info = nodes[n.name];
    std::vector<Output> ins;
    for (auto& p : info.inputs) ins.emplace_back(p.first, p.second);
    std::vector<Output> gins;
    STF_RETURN_IF_ERROR(rit->second(scope, info, ins, gout, &gins));
    for (size_t i = 0; i < gins.size() && i < info.inputs.size(); ++i) {
      if (!gins[i].valid()) continue;
      std::string key =
          key_of(info.inputs[i].first, info.inputs[i].second);
      auto a = accum.find(key);
      if (a == accum.end()) {
        accum[key] = gins[i];
      } else {
        a->second = ops::Add(scope, a->second, gins[i]);
      }
    }
  }
  grad_outputs->clear();
  for (auto& in : inputs) {
    auto a = accum.find(key_of(in.node, in.index));
    if (a == accum.end())
      return errors::InvalidArgument("No gradient flows to ", in.name());
    grad_outputs->push_back(a->second);
  }
  return scope.status();
}

Status AddSymbolicGradients(const Scope& scope,
                            const std::vector<Output>& outputs,
                            const std::vector<Output>& inputs,
                            std::vector<Output>* grad_outputs) {
  return AddSymbolicGradients(scope, outputs, inputs, {}, grad_outputs);
}

}  // namespace cc
}  // namespace stf
