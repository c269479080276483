// Session-level graph optimizations: common-subexpression elimination and
// constant folding (capability analogs of the reference's
// core/graph/optimizer_cse.cc and common_runtime/constant_folding.cc,
// redesigned over this framework's Graph/NodeDef structures: folding runs
// the real CPU kernels in place instead of spinning up a GraphRunner).
#include <map>
#include <set>
#include <string>
#include <vector>

#include "framework/op_kernel.h"
#include "graph/graph.h"

namespace stf {

namespace {

// Ops that must never be folded or merged even though they look pure.
bool Blacklisted(const Node* n) {
  const std::string& op = n->op();
  return n->IsControlFlow() || n->IsSend() || n->IsRecv() ||
         op == "Placeholder" ||
         op == "PlaceholderWithDefault" || op == "NoOp" ||
         StrStartsWith(op, "Rccl") || StrStartsWith(op, "Random") ||
         StrStartsWith(op, "TruncatedNormal") || op == "PyFunc";
}

bool IsPure(const Node* n) {
  if (Blacklisted(n)) return false;
  if (n->op_def && n->op_def->is_stateful) return false;
  for (bool r : n->out_is_ref)
    if (r) return false;
  for (DataType t : n->in_types)
    if (t == DT_STRING) return false;  // handles masquerade as strings
  // A node reading a ref output (variable snapshot) is ordered against
  // writers only by control edges on THIS node; merging two reads (or
  // folding one) could move a read across an assign. Leave them alone.
  for (const Edge* e : n->in_edges)
    if (!e->IsControl() && e->src_output < (int)e->src->out_is_ref.size() &&
        e->src->out_is_ref[e->src_output])
      return false;
  return true;
}

std::string AttrFingerprint(const NodeDef& def) {
  pb::Writer w;
  for (const auto& kv : def.attr) {  // std::map: deterministic order
    pb::Writer av;
    kv.second.Serialize(&av);
    w.PutString(1, kv.first);
    w.PutString(2, av.buf());
  }
  return w.buf();
}

// ------------------------------- CSE ---------------------------------------
int EliminateCommonSubexpressions(Graph* g,
                                  const std::set<std::string>& preserve) {
  std::vector<Node*> order;
  if (!TopologicalOrder(*g, &order).ok()) return 0;
  std::map<std::string, Node*> canonical;
  int merged = 0;
  for (Node* n : order) {
    if (!IsPure(n) || preserve.count(n->name())) continue;
    // signature: op + attrs + device + ordered data inputs + control set
    std::string sig = n->op() + '\0' + n->def.device + '\0' +
                      AttrFingerprint(n->def);
    std::vector<std::string> ctrl;
    std::map<int, std::string> data;
    bool ok = true;
    for (const Edge* e : n->in_edges) {
      if (e->IsControl()) {
        ctrl.push_back(e->src->name());
      } else {
        data[e->dst_input] = e->src->name() + ':' +
                             std::to_string(e->src_output);
      }
    }
    for (auto& kv : data) sig += '\1' + kv.second;
    std::sort(ctrl.begin(), ctrl.end());
    for (auto& c : ctrl) sig += '\2' + c;
    (void)ok;
    auto it = canonical.find(sig);
    if (it == canonical.end()) {
      canonical.emplace(std::move(sig), n);
      continue;
    }
    Node* rep = it->second;
    // redirect all out edges of n to rep
    std::vector<Edge*> outs(n->out_edges.begin(), n->out_edges.end());
    for (Edge* e : outs) {
      Node* dst = e->dst;
      int di = e->dst_input;
      int so = e->src_output;
      g->RemoveEdge(e);
      if (so < 0)
        g->AddControlEdge(rep, dst);
      else
        g->AddEdge(rep, so, dst, di);
    }
    g->RemoveNode(n);
    ++merged;
  }
  return merged;
}

// -------------------------- constant folding --------------------------------
constexpr int64_t kMaxFoldElements = 1 << 20;

bool AllInputsConst(const Node* n) {
  bool any = false;
  for (const Edge* e : n->in_edges) {
    if (e->IsControl()) return false;  // preserve control semantics
    if (!e->src->IsConstant()) return false;
    any = true;
  }
  return any;
}

int FoldConstants(Graph* g, Device* cpu,
                  const std::set<std::string>& preserve) {
  int folded_total = 0;
  bool changed = true;
  while (changed) {
    changed = false;
    std::vector<Node*> order;
    if (!TopologicalOrder(*g, &order).ok()) return folded_total;
    for (Node* n : order) {
      if (!IsPure(n) || n->num_outputs() != 1 || !AllInputsConst(n) ||
          preserve.count(n->name()))
        continue;
      if (!KernelRegistry::Global()->HasKernel(n->def, "CPU")) continue;
      // materialize const inputs
      std::vector<Tensor> inputs(n->num_inputs());
      bool ready = true;
      for (const Edge* e : n->in_edges) {
        auto it = e->src->def.attr.find("value");
        if (it == e->src->def.attr.end() || it->second.kind != 'e') {
          ready = false;
          break;
        }
        Tensor t;
        if (!Tensor::FromProto(it->second.tensor, &t).ok()) {
          ready = false;
          break;
        }
        inputs[e->dst_input] = t;
      }
      if (!ready) continue;
      std::unique_ptr<OpKernel> kernel;
      if (!CreateOpKernel("CPU", cpu, n->def, &kernel).ok()) continue;
      OpKernelContext ctx(kernel.get(), cpu, std::move(inputs));
      kernel->Compute(&ctx);
      if (!ctx.status().ok()) continue;
      const Tensor& result = ctx.output(0);
      if (!result.IsInitialized() ||
          result.NumElements() > kMaxFoldElements ||
          result.dtype() == DT_STRING)
        continue;
      // build replacement Const
      NodeDef cdef;
      cdef.name = n->name() + "/_folded";
      cdef.op = "Const";
      cdef.device = n->def.device;
      AttrValue dt;
      dt.kind = 't';
      dt.type = result.dtype();
      cdef.attr["dtype"] = dt;
      AttrValue val;
      val.kind = 'e';
      result.AsProto(&val.tensor);
      cdef.attr["value"] = val;
      Node* cnode = nullptr;
      if (!g->AddNode(cdef, &cnode).ok()) continue;
      std::vector<Edge*> outs(n->out_edges.begin(), n->out_edges.end());
      for (Edge* e : outs) {
        Node* dst = e->dst;
        int di = e->dst_input;
        int so = e->src_output;
        g->RemoveEdge(e);
        if (so < 0)
          g->AddControlEdge(cnode, dst);
        else
          g->AddEdge(cnode, 0, dst, di);
      }
      g->RemoveNode(n);
      ++folded_total;
      changed = true;
    }
  }
  return folded_total;
}

}  // namespace

// Runs CSE + constant folding to fixpoint (preserving feed/fetch/target
// nodes). Returns number of nodes removed.
int OptimizeGraph(Graph* g, Device* cpu,
                  const std::set<std::string>& preserve) {
  static const bool disabled = getenv("STF_NO_GRAPH_OPT") != nullptr;
  if (disabled) return 0;
  int total = 0;
  for (int pass = 0; pass < 3; ++pass) {
    int changed = EliminateCommonSubexpressions(g, preserve);
    changed += FoldConstants(g, cpu, preserve);
    total += changed;
    if (!changed) break;
  }
  return total;
}

}  // namespace stf
