#include "graph/graph.h"

#include <algorithm>
#include <deque>

namespace stf {

Status Graph::AddNode(const NodeDef& def, Node** out) {
  const OpDef* op_def = OpRegistry::Global()->LookUp(def.op);
  if (!op_def) return errors::NotFound("Op not registered: ", def.op);
  if (by_name_.count(def.name))
    return errors::AlreadyExists("Duplicate node name: ", def.name);
  auto n = std::make_unique<Node>();
  n->id = (int)nodes_.size();
  n->def = def;
  AddDefaultsToNodeDef(*op_def, &n->def);
  n->op_def = op_def;
  STF_RETURN_IF_ERROR(
      InOutTypesForNode(n->def, *op_def, &n->in_types, &n->out_types,
                        &n->out_is_ref));
  Node* raw = n.get();
  by_name_[def.name] = raw;
  alive_.push_back(raw);
  nodes_.push_back(std::move(n));
  if (out) *out = raw;
  return Status::OK();
}

Edge* Graph::AddEdge(Node* src, int src_output, Node* dst, int dst_input) {
  auto e = std::make_unique<Edge>();
  e->src = src;
  e->src_output = src_output;
  e->dst = dst;
  e->dst_input = dst_input;
  Edge* raw = e.get();
  src->out_edges.push_back(raw);
  dst->in_edges.push_back(raw);
  edges_.push_back(std::move(e));
  return raw;
}

Edge* Graph::AddControlEdge(Node* src, Node* dst) {
  return AddEdge(src, -1, dst, -1);
}

void Graph::RemoveEdge(Edge* e) {
  auto& so = e->src->out_edges;
  so.erase(std::remove(so.begin(), so.end(), e), so.end());
  auto& di = e->dst->in_edges;
  di.erase(std::remove(di.begin(), di.end(), e), di.end());
  e->src = e->dst = nullptr;  // tombstone; storage freed with graph
}

void Graph::RemoveNode(Node* n) {
  std::vector<Edge*> edges = n->in_edges;
  edges.insert(edges.end(), n->out_edges.begin(), n->out_edges.end());
  for (Edge* e : edges)
    if (e->src) RemoveEdge(e);
  by_name_.erase(n->name());
  alive_.erase(std::remove(alive_.begin(), alive_.end(), n), alive_.end());
}

void Graph::ToGraphDef(GraphDef* out) const {
  out->node.clear();
  for (Node* n : alive_) {
    NodeDef d = n->def;
    d.input.clear();
    std::vector<const Edge*> data(n->num_inputs(), nullptr);
    std::vector<const Edge*> ctrl;
    for (auto* e : n->in_edges) {
      if (e->IsControl()) ctrl.push_back(e);
      else data[e->dst_input] = e;
    }
    for (auto* e : data) {
      if (!e) continue;
      d.input.push_back(e->src_output == 0
                            ? e->src->name()
                            : e->src->name() + ":" + std::to_string(e->src_output));
    }
    for (auto* e : ctrl) d.input.push_back("^" + e->src->name());
    out->node.push_back(std::move(d));
  }
}

Status ConvertGraphDefToGraph(const GraphDef& gdef, Graph* g) {
  std::vector<Node*> nodes;
  nodes.reserve(gdef.node.size());
  for (auto& nd : gdef.node) {
    Node* n;
    STF_RETURN_IF_ERROR(g->AddNode(nd, &n));
    nodes.push_back(n);
  }
  for (size_t i = 0; i < gdef.node.size(); ++i) {
    const NodeDef& nd = gdef.node[i];
    Node* dst = nodes[i];
    int slot = 0;
    for (auto& in : nd.input) {
      if (!in.empty() && in[0] == '^') {
        Node* src = g->FindNode(in.substr(1));
        if (!src)
          return errors::InvalidArgument("Unknown control input ", in,
                                         " of node ", nd.name);
        g->AddControlEdge(src, dst);
      } else {
        std::string name = in;
        int port = 0;
        auto colon = in.rfind(':');
        if (colon != std::string::npos &&
            in.find_first_not_of("0123456789", colon + 1) == std::string::npos) {
          name = in.substr(0, colon);
          port = atoi(in.c_str() + colon + 1);
        }
        Node* src = g->FindNode(name);
        if (!src)
          return errors::InvalidArgument("Unknown input ", in, " of node ",
                                         nd.name);
        if (port >= src->num_outputs())
          return errors::InvalidArgument("Input ", in, " of node ", nd.name,
                                         ": port out of range");
        g->AddEdge(src, port, dst, slot);
        ++slot;
      }
    }
    if (slot != dst->num_inputs())
      return errors::InvalidArgument("Node ", nd.name, " (", nd.op, ") has ",
                                     slot, " data inputs, expected ",
                                     dst->num_inputs());
  }
  return Status::OK();
}

Status TopologicalOrder(const Graph& g, std::vector<Node*>* order) {
  std::map<Node*, int> pending;
  std::deque<Node*> ready;
  for (Node* n : g.nodes()) {
    int cnt = 0;
    for (auto* e : n->in_edges)
      if (!e->src->IsNextIteration()) ++cnt;
    pending[n] = cnt;
    if (cnt == 0) ready.push_back(n);
  }
  order->clear();
  while (!ready.empty()) {
    Node* n = ready.front();
    ready.pop_front();
    order->push_back(n);
    for (auto* e : n->out_edges) {
      if (n->IsNextIteration()) continue;
      if (--pending[e->dst] == 0) ready.push_back(e->dst);
    }
  }
  if (order->size() != g.nodes().size())
    return errors::InvalidArgument("Graph has a cycle (outside while-loops)");
  return Status::OK();
}

}  // namespace stf
