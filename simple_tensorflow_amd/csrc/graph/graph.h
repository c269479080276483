// In-memory Graph/Node/Edge + GraphDef construction.
// Capability analog of the reference's core/graph/graph.h +
// graph_constructor.cc, compacted: no source/sink pseudo-nodes; edges are
// stored per-node.
#pragma once

#include <map>
#include <memory>
#include <set>
#include <string>
#include <vector>

#include "core/base.h"
#include "core/protos.h"
#include "framework/op.h"

namespace stf {

class Node;

struct Edge {
  Node* src = nullptr;
  int src_output = 0;  // -1 for control edges
  Node* dst = nullptr;
  int dst_input = 0;  // -1 for control edges
  bool IsControl() const { return src_output < 0; }
};

class Node {
 public:
  int id = -1;
  NodeDef def;
  const OpDef* op_def = nullptr;
  std::vector<DataType> in_types;
  std::vector<DataType> out_types;
  std::vector<bool> out_is_ref;
  std::vector<Edge*> in_edges;   // includes control edges
  std::vector<Edge*> out_edges;  // includes control edges
  std::string assigned_device;   // canonical "CPU:0"/"GPU:1" after placement

  const std::string& name() const { return def.name; }
  const std::string& op() const { return def.op; }
  int num_inputs() const { return (int)in_types.size(); }
  int num_outputs() const { return (int)out_types.size(); }

  bool IsSwitch() const { return def.op == "Switch" || def.op == "RefSwitch"; }
  bool IsMerge() const { return def.op == "Merge" || def.op == "RefMerge"; }
  bool IsEnter() const { return def.op == "Enter" || def.op == "RefEnter"; }
  bool IsExit() const { return def.op == "Exit" || def.op == "RefExit"; }
  bool IsNextIteration() const {
    return def.op == "NextIteration" || def.op == "RefNextIteration";
  }
  bool IsConstant() const { return def.op == "Const"; }
  bool IsSend() const { return def.op == "_Send" || def.op == "_HostSend"; }
  bool IsRecv() const { return def.op == "_Recv" || def.op == "_HostRecv"; }
  bool IsControlFlow() const {
    return IsSwitch() || IsMerge() || IsEnter() || IsExit() || IsNextIteration();
  }

  // Finds the data edge feeding input slot i (nullptr if missing).
  const Edge* input_edge(int i) const {
    for (auto* e : in_edges)
      if (!e->IsControl() && e->dst_input == i) return e;
    return nullptr;
  }
};

class Graph {
 public:
  Graph() {}

  // Adds a node for `def` (resolving the OpDef and types). Does not connect
  // inputs — use AddEdge, or build via GraphConstructor.
  Status AddNode(const NodeDef& def, Node** out);
  Edge* AddEdge(Node* src, int src_output, Node* dst, int dst_input);
  Edge* AddControlEdge(Node* src, Node* dst);
  void RemoveNode(Node* n);
  void RemoveEdge(Edge* e);

  Node* FindNode(const std::string& name) const {
    auto it = by_name_.find(name);
    return it == by_name_.end() ? nullptr : it->second;
  }

  const std::vector<Node*>& nodes() const { return alive_; }
  int num_node_ids() const { return (int)nodes_.size(); }
  Node* FindNodeId(int id) const { return nodes_[id].get(); }

  // Re-serialize to GraphDef (inputs rebuilt from edges).
  void ToGraphDef(GraphDef* out) const;

 private:
  std::vector<std::unique_ptr<Node>> nodes_;
  std::vector<std::unique_ptr<Edge>> edges_;
  std::vector<Node*> alive_;
  std::map<std::string, Node*> by_name_;
};

// Builds `g` from `gdef`: creates all nodes then resolves input strings
// ("name", "name:k", "^name").
Status ConvertGraphDefToGraph(const GraphDef& gdef, Graph* g);

// Topological order (ignores NextIteration back-edges). Returns error on
// cycles that are not while-loop back-edges.
Status TopologicalOrder(const Graph& g, std::vector<Node*>* order);

class Device;

// Session-level graph optimization: common-subexpression elimination +
// constant folding to fixpoint (graph/optimizer.cc). Nodes named in
// `preserve` (feeds/fetches/targets) are never removed. `cpu` is the host
// device used to execute foldable subgraphs. Returns nodes eliminated.
int OptimizeGraph(Graph* g, Device* cpu, const std::set<std::string>& preserve);

}  // namespace stf
