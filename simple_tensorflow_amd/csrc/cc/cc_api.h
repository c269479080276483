// C++ graph-construction API: Scope + typed op builders + ClientSession +
// AddSymbolicGradients (capability analog of the reference cc/ layer:
// framework/scope.h, client/client_session.h, framework/gradients.cc and the
// cc_op_gen-generated op wrappers — here a compact hand-written set over the
// same OpRegistry).
#pragma once

#include <map>
#include <memory>
#include <string>
#include <vector>

#include "core/protos.h"
#include "core/tensor.h"

namespace stf {

class DirectSession;

namespace cc {

struct Output {
  std::string node;
  int index = 0;
  Output() {}
  Output(std::string n, int i = 0) : node(std::move(n)), index(i) {}
  std::string name() const {
    return index == 0 ? node : node + ":" + std::to_string(index);
  }
  bool valid() const { return !node.empty(); }
};

struct GraphState {
  GraphDef gdef;
  std::map<std::string, int> names_used;
  std::map<std::string, const NodeDef*> by_name;
  Status status;
};

class Scope {
 public:
  static Scope NewRootScope();
  Scope NewSubScope(const std::string& child) const;

  // Generic builder: resolves output arity from the OpDef; returns output 0.
  Output AddOp(const std::string& op_type, const std::string& name,
               const std::vector<Output>& inputs,
               const std::map<std::string, AttrValue>& attrs = {}) const;

  GraphDef ToGraphDef() const { return state_->gdef; }
  Status status() const { return state_->status; }
  GraphState* state() const { return state_.get(); }
  std::string UniqueName(const std::string& base) const;

 private:
  Scope(std::shared_ptr<GraphState> s, std::string prefix)
      : state_(std::move(s)), prefix_(std::move(prefix)) {}
  std::shared_ptr<GraphState> state_;
  std::string prefix_;
};

// ---- typed op builders (subset of reference cc/ops/) ----
namespace ops {
Output Const(const Scope& s, const Tensor& value,
             const std::string& name = "Const");
Output Const(const Scope& s, float value);
Output Const(const Scope& s, const std::vector<float>& value,
             const std::vector<int64_t>& shape);
Output Placeholder(const Scope& s, DataType dtype,
                   const std::string& name = "Placeholder");
Output Add(const Scope& s, Output a, Output b);
Output Sub(const Scope& s, Output a, Output b);
Output Mul(const Scope& s, Output a, Output b);
Output Div(const Scope& s, Output a, Output b);
Output Neg(const Scope& s, Output a);
Output Square(const Scope& s, Output a);
Output Relu(const Scope& s, Output a);
Output Tanh(const Scope& s, Output a);
Output Sigmoid(const Scope& s, Output a);
Output Exp(const Scope& s, Output a);
Output Identity(const Scope& s, Output a);
Output MatMul(const Scope& s, Output a, Output b, bool transpose_a = false,
              bool transpose_b = false);
Output Softmax(const Scope& s, Output logits);
Output ReduceSum(const Scope& s, Output a, const std::vector<int>& axes);
Output ReduceMean(const Scope& s, Output a, const std::vector<int>& axes);
Output Reshape(const Scope& s, Output a, const std::vector<int64_t>& shape);
Output OnesLike(const Scope& s, Output a);
Output ZerosLike(const Scope& s, Output a);
}  // namespace ops

// ---- C++ autodiff (reference cc/framework/gradients.cc
// AddSymbolicGradients + grad_op_registry.h) ----
Status AddSymbolicGradients(const Scope& scope,
                            const std::vector<Output>& outputs,
                            const std::vector<Output>& inputs,
                            const std::vector<Output>& grad_inputs,
                            std::vector<Output>* grad_outputs);
Status AddSymbolicGradients(const Scope& scope,
                            const std::vector<Output>& outputs,
                            const std::vector<Output>& inputs,
                            std::vector<Output>* grad_outputs);

// ---- ClientSession (reference cc/client/client_session.h) ----
class ClientSession {
 public:
  explicit ClientSession(const Scope& scope);
  ~ClientSession();
  Status Run(const std::vector<Output>& fetches,
             std::vector<Tensor>* outputs);
  Status Run(const std::vector<std::pair<Output, Tensor>>& feeds,
             const std::vector<Output>& fetches,
             std::vector<Tensor>* outputs);
  Status Run(const std::vector<std::pair<Output, Tensor>>& feeds,
             const std::vector<Output>& fetches,
             const std::vector<Output>& targets, std::vector<Tensor>* outputs);

 private:
  const Scope scope_;
  std::unique_ptr<DirectSession> sess_;
  size_t created_nodes_ = 0;
};

}  // namespace cc
}  // namespace stf
