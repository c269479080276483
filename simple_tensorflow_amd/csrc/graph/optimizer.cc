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
#include "kernels/fused_ew.h"

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


// ------------------------- elementwise fusion -------------------------------
// Scoped fusion pass (SURVEY §7.10 slot): collapses a single-root DAG of
// elementwise ops (unaries + binaries whose every operand derives from the
// same root tensor or is a scalar constant) into ONE _FusedElementwise node
// executing a register-resident bytecode program. Single-root + scalar-only
// side inputs makes the no-broadcast guarantee provable without shape
// inference: every value in the group has exactly the root's shape.
namespace {

int FuseOpcode(const std::string& op) {
  using namespace fused_ew;
  if (op == "Relu") return kRelu;
  if (op == "Relu6") return kRelu6;
  if (op == "Sigmoid") return kSigmoid;
  if (op == "Tanh") return kTanh;
  if (op == "Exp") return kExp;
  if (op == "Log") return kLog;
  if (op == "Log1p") return kLog1p;
  if (op == "Neg") return kNeg;
  if (op == "Sqrt") return kSqrt;
  if (op == "Rsqrt") return kRsqrt;
  if (op == "Square") return kSquare;
  if (op == "Abs") return kAbs;
  if (op == "Softplus") return kSoftplus;
  if (op == "Sign") return kSign;
  if (op == "Floor") return kFloor;
  if (op == "Reciprocal") return kReciprocal;
  if (op == "Add") return kAdd;
  if (op == "Sub") return kSub;
  if (op == "Mul") return kMul;
  if (op == "Div" || op == "RealDiv") return kDiv;
  if (op == "Maximum") return kMaximum;
  if (op == "Minimum") return kMinimum;
  if (op == "SquaredDifference") return kSquaredDifference;
  if (op == "Pow") return kPow;
  return 0;
}

bool IsScalarConst(const Node* n) {
  if (!n->IsConstant()) return false;
  auto it = n->def.attr.find("value");
  if (it == n->def.attr.end() || it->second.kind != 'e') return false;
  const TensorShapeProto& sh = it->second.tensor.tensor_shape;
  int64_t elems = 1;
  for (auto& d : sh.dim) elems *= d.size;
  return elems == 1;
}

bool FusableDtype(const Node* n) {
  if (n->out_types.size() != 1) return false;
  DataType t = n->out_types[0];
  if (t != DT_FLOAT && t != DT_BFLOAT16) return false;
  for (DataType it : n->in_types)
    if (it != t) return false;
  return true;
}

int FuseElementwise(Graph* g, const std::set<std::string>& preserve) {
  std::vector<Node*> order;
  if (!TopologicalOrder(*g, &order).ok()) return 0;
  // root(n): the unique non-scalar, non-fusable source this node's value
  // derives from ("" port key), or nullptr if mixed roots / not fusable.
  std::map<Node*, std::pair<Node*, int>> root;  // node -> (root node, port)
  std::map<Node*, int> opcode;
  for (Node* n : order) {
    if (preserve.count(n->name())) continue;
    int oc = FuseOpcode(n->op());
    if (!oc || !FusableDtype(n)) continue;
    bool has_control = false;
    for (const Edge* e : n->in_edges)
      if (e->IsControl()) has_control = true;
    if (has_control) continue;
    Node* r = nullptr;
    int rp = 0;
    bool ok = true;
    for (const Edge* e : n->in_edges) {
      Node* src = e->src;
      if (IsScalarConst(src)) continue;
      Node* cand;
      int cand_p;
      auto it = root.find(src);
      if (it != root.end()) {
        cand = it->second.first;
        cand_p = it->second.second;
      } else {
        cand = src;
        cand_p = e->src_output;
      }
      if (!r) {
        r = cand;
        rp = cand_p;
      } else if (r != cand || rp != cand_p) {
        ok = false;
        break;
      }
    }
    if (ok && r) {
      root[n] = {r, rp};
      opcode[n] = oc;
    }
  }
  // group members by root
  std::map<std::pair<Node*, int>, std::vector<Node*>> groups;
  for (Node* n : order) {
    auto it = root.find(n);
    if (it != root.end()) groups[it->second].push_back(n);  // topo-ordered
  }
  int fused_total = 0;
  for (auto& kv : groups) {
    const std::vector<Node*>& members = kv.second;
    std::set<Node*> mem(members.begin(), members.end());
    // Every member with an external consumer is an exit; each exit gets its
    // own fused node over its backward closure (shared interiors are
    // recomputed inside each program — register math is far cheaper than
    // the HBM round-trips of materializing them). Originals are removed
    // afterwards when nothing external reads them anymore.
    std::vector<Node*> exits;
    for (Node* n : members) {
      bool escapes = n->out_edges.empty();
      for (const Edge* e : n->out_edges)
        if (!mem.count(e->dst)) escapes = true;
      if (escapes) exits.push_back(n);
    }
    for (Node* exit_node : exits) {
    // closure feeding the exit
    std::vector<Node*> chain;
    std::set<Node*> keep;
    std::vector<Node*> stack = {exit_node};
    while (!stack.empty()) {
      Node* n = stack.back();
      stack.pop_back();
      if (!keep.insert(n).second) continue;
      for (const Edge* e : n->in_edges)
        if (mem.count(e->src)) stack.push_back(e->src);
    }
    for (Node* n : members)
      if (keep.count(n)) chain.push_back(n);  // stays topo-ordered
    if ((int)chain.size() < 2 ||
        (int)chain.size() > fused_ew::kMaxInstr)
      continue;
    // collect inputs: slot 0 = root, then scalar consts in first-use order
    Node* rnode = kv.first.first;
    int rport = kv.first.second;
    std::vector<std::pair<Node*, int>> inputs = {{rnode, rport}};
    std::map<Node*, int> slot_of_input;  // scalar const node -> slot
    std::map<Node*, int> slot_of;        // chain node -> slot
    auto input_slot = [&](Node* src, int port) -> int {
      if (src == rnode) return 0;
      auto it = slot_of_input.find(src);
      if (it != slot_of_input.end()) return it->second;
      int s = (int)inputs.size();
      inputs.push_back({src, port});
      slot_of_input[src] = s;
      return s;
    };
    bool ok = true;
    // first pass: count scalar inputs
    for (Node* n : chain) {
      for (const Edge* e : n->in_edges) {
        if (mem.count(e->src) && keep.count(e->src)) continue;
        if (IsScalarConst(e->src)) {
          input_slot(e->src, e->src_output);
        } else if (!(e->src == rnode && e->src_output == rport)) {
          ok = false;  // should not happen per grouping, stay safe
        }
      }
      if (!ok) break;
    }
    if (!ok || (int)inputs.size() > fused_ew::kMaxInputs) continue;
    int n_in = (int)inputs.size();
    // program
    std::vector<int64_t> prog;
    int next_slot = n_in;
    for (Node* n : chain) {
      int srcs[2] = {0, 0};
      int si = 0;
      for (const Edge* e : n->in_edges) {
        if (e->IsControl()) continue;
        int slot;
        if (keep.count(e->src))
          slot = slot_of[e->src];
        else if (e->src == rnode && e->src_output == rport)
          slot = 0;
        else
          slot = input_slot(e->src, e->src_output);
        if (si < 2) srcs[e->dst_input < 2 ? e->dst_input : si] = slot;
        ++si;
      }
      prog.push_back(fused_ew::Pack(opcode[n], srcs[0], srcs[1]));
      slot_of[n] = next_slot++;
    }
    // build the fused node
    NodeDef fdef;
    fdef.name = exit_node->name() + "/_fused";
    fdef.op = "_FusedElementwise";
    fdef.device = exit_node->def.device;
    AttrValue t;
    t.kind = 't';
    t.type = exit_node->out_types[0];
    fdef.attr["T"] = t;
    AttrValue nattr;
    nattr.kind = 'i';
    nattr.i = n_in;
    fdef.attr["N"] = nattr;
    AttrValue pattr;
    pattr.kind = 'l';
    pattr.list.i = prog;
    fdef.attr["program"] = pattr;
    Node* fnode = nullptr;
    if (!g->AddNode(fdef, &fnode).ok()) continue;
    for (int i = 0; i < n_in; ++i)
      g->AddEdge(inputs[i].first, inputs[i].second, fnode, i);
    // rewire exit's external consumers
    std::vector<Edge*> outs(exit_node->out_edges.begin(),
                            exit_node->out_edges.end());
    for (Edge* e : outs) {
      if (mem.count(e->dst)) continue;  // stays for other exits' closures
      Node* dst = e->dst;
      int di = e->dst_input;
      int so = e->src_output;
      g->RemoveEdge(e);
      if (so < 0)
        g->AddControlEdge(fnode, dst);
      else
        g->AddEdge(fnode, 0, dst, di);
    }
    fused_total += (int)chain.size();
    }  // exits
    // drop members that nothing external consumes anymore (reverse topo)
    for (auto it = members.rbegin(); it != members.rend(); ++it) {
      bool busy = false;
      for (const Edge* e : (*it)->out_edges)
        if (!mem.count(e->dst)) busy = true;
      if (!busy) g->RemoveNode(*it);
    }
  }
  return fused_total;
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
  static const bool no_fusion = getenv("STF_NO_FUSION") != nullptr;
  if (!no_fusion) total += FuseElementwise(g, preserve);
  return total;
}

}  // namespace stf
