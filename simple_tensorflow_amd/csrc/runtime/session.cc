#include "runtime/session.h"

#include "gpu/gpu_tracer.h"

#include <algorithm>
#include <atomic>
#include <chrono>
#include <condition_variable>
#include <set>
#include <thread>

#include "graph/graph.h"

namespace stf {

// Implemented in gpu/gpu_device.cc; adds one Device per visible HIP GPU.
void AddGpuDevices(DeviceMgr* mgr);
// Implemented in kernels/resource_mgr.cc.
void* NewResourceMgr();
void DeleteResourceMgr(void*);

namespace {

// "name" or "name:3" -> (name, port)
std::pair<std::string, int> ParseTensorName(const std::string& s) {
  auto colon = s.rfind(':');
  if (colon != std::string::npos &&
      s.find_first_not_of("0123456789", colon + 1) == std::string::npos &&
      colon + 1 < s.size())
    return {s.substr(0, colon), atoi(s.c_str() + colon + 1)};
  return {s, 0};
}

NodeDef MakeSendRecv(const std::string& op, const std::string& name,
                     DataType dtype, const std::string& tensor_name,
                     const std::string& send_dev, const std::string& recv_dev) {
  NodeDef d;
  d.name = name;
  d.op = op;
  d.attr[op == "_Recv" ? "tensor_type" : "T"] = AttrValue::Type(dtype);
  d.attr["tensor_name"] = AttrValue::S(tensor_name);
  d.attr["send_device"] = AttrValue::S(send_dev);
  d.attr["recv_device"] = AttrValue::S(recv_dev);
  return d;
}

}  // namespace

DirectSession::DirectSession(bool force_cpu_only, int num_threads) {
  devices_.AddDevice(std::make_unique<ThreadPoolDevice>("/cpu:0"));
  if (!force_cpu_only) AddGpuDevices(&devices_);
  int n = num_threads > 0 ? num_threads
                          : std::max(4u, std::thread::hardware_concurrency());
  pool_ = std::make_unique<ThreadPool>(n);
  resource_mgr_ = NewResourceMgr();
}

DirectSession::~DirectSession() { DeleteResourceMgr(resource_mgr_); }

Status DirectSession::Create(const GraphDef& def) {
  std::lock_guard<std::mutex> l(mu_);
  graph_def_ = def;
  executors_.clear();
  return Status::OK();
}

Status DirectSession::Extend(const GraphDef& def) {
  std::lock_guard<std::mutex> l(mu_);
  for (auto& n : def.node) graph_def_.node.push_back(n);
  executors_.clear();
  return Status::OK();
}

Status DirectSession::GetOrCreateExecutors(
    const std::vector<std::string>& feeds,
    const std::vector<std::string>& fetches,
    const std::vector<std::string>& targets, ExecutorsAndKeys** out) {
  std::string key;
  for (auto& f : feeds) key += f + ",";
  key += "|";
  for (auto& f : fetches) key += f + ",";
  key += "|";
  for (auto& t : targets) key += t + ",";
  {
    std::lock_guard<std::mutex> l(mu_);
    auto it = executors_.find(key);
    if (it != executors_.end()) {
      *out = it->second.get();
      return Status::OK();
    }
  }
  std::unique_ptr<ExecutorsAndKeys> ek;
  STF_RETURN_IF_ERROR(BuildExecutors(feeds, fetches, targets, &ek));
  std::lock_guard<std::mutex> l(mu_);
  auto& slot = executors_[key];
  if (!slot) slot = std::move(ek);
  *out = slot.get();
  return Status::OK();
}

Status DirectSession::BuildExecutors(const std::vector<std::string>& feeds,
                                     const std::vector<std::string>& fetches,
                                     const std::vector<std::string>& targets,
                                     std::unique_ptr<ExecutorsAndKeys>* out) {
  GraphDef gdef_copy;
  {
    std::lock_guard<std::mutex> l(mu_);
    gdef_copy = graph_def_;
  }
  auto graph = std::make_unique<Graph>();
  STF_RETURN_IF_ERROR(ConvertGraphDefToGraph(gdef_copy, graph.get()));
  auto ek = std::make_unique<ExecutorsAndKeys>();

  // ---- 0. Graph optimization (CSE + constant folding) on the pruned-to-be
  // graph. Feeds/fetches/targets must survive by name. Capability analog of
  // the reference's OptimizeCSE + DoConstantFolding pre-passes.
  {
    std::set<std::string> preserve;
    for (auto& f : feeds) preserve.insert(ParseTensorName(f).first);
    for (auto& f : fetches) preserve.insert(ParseTensorName(f).first);
    for (auto& t : targets) preserve.insert(ParseTensorName(t).first);
    OptimizeGraph(graph.get(), devices_.LookUp("/cpu:0"), preserve);
  }

  // ---- 1. Placement (before rewrite so feed recvs inherit devices). ----
  std::vector<Node*> order;
  STF_RETURN_IF_ERROR(TopologicalOrder(*graph, &order));
  Device* cpu = devices_.LookUp("/cpu:0");
  Device* gpu0 = nullptr;
  for (auto& d : devices_.devices())
    if (d->is_gpu() && !gpu0) gpu0 = d.get();

  auto kernel_available = [&](Node* n, Device* d) {
    if (n->IsControlFlow() || n->IsSend() || n->IsRecv() || n->op() == "NoOp")
      return true;
    return KernelRegistry::Global()->HasKernel(n->def, d->device_type());
  };
  for (Node* n : order) {
    std::string req = CanonicalDevice(n->def.device);
    Device* dev = nullptr;
    // Shape-carrying int32 constants live on the host: GPU kernels take
    // shape/axes/perm args in host memory, and a device-resident int32 Const
    // would force a per-step d2h sync at every consumer.
    if (req.empty() && n->op() == "Const") {
      DataType dt;
      if (GetAttrType(n->def, "dtype", &dt) && dt == DT_INT32) {
        n->assigned_device = CanonicalDevice(cpu->name());
        continue;
      }
    }
    // String tensors are host-only (variable-length payloads have no device
    // representation). TensorArray-style ops carry string HANDLES, but their
    // GPU kernels declare them HostMemory — those stay GPU-eligible so
    // while-loop frames remain single-partition (the executor has no
    // cross-partition control loops). Pin to CPU only when GPU placement
    // would put a string payload in device memory.
    {
      bool touches_string = false;
      for (auto t : n->out_types)
        if (t == DT_STRING) touches_string = true;
      for (auto t : n->in_types)
        if (t == DT_STRING) touches_string = true;
      // Control-flow ops (Enter/Exit/Switch/Merge/NextIteration) forward
      // tensors without allocating: a host-resident string handle passes
      // through them untouched, and pinning them would split a while frame
      // across partitions (deadlock). Exempt them.
      if (touches_string && !n->IsControlFlow()) {
        bool gpu_ok = false;
        const KernelDef* kd =
            gpu0 ? KernelRegistry::Global()->Find(n->def, "GPU") : nullptr;
        const OpDef* od = OpRegistry::Global()->LookUp(n->op());
        if (kd && od) {
          // Only fixed-arity signatures are checked arg-by-arg; ops with
          // list-expanded string args stay conservative (CPU).
          gpu_ok = od->output_arg.size() == n->out_types.size() &&
                   od->input_arg.size() == n->in_types.size();
          if (gpu_ok)
            for (size_t i = 0; i < n->out_types.size(); ++i)
              if (n->out_types[i] == DT_STRING &&
                  !kd->host_memory.count(od->output_arg[i].name))
                gpu_ok = false;
          if (gpu_ok)
            for (size_t i = 0; i < n->in_types.size(); ++i)
              if (n->in_types[i] == DT_STRING &&
                  !kd->host_memory.count(od->input_arg[i].name))
                gpu_ok = false;
        }
        if (!gpu_ok) {
          n->assigned_device = CanonicalDevice(cpu->name());
          continue;
        }
      }
    }
    if (!req.empty()) {
      dev = devices_.LookUp(req);
      if (!dev && StrStartsWith(req, "GPU") && gpu0) dev = gpu0;
      if (!dev) dev = cpu;  // soft placement
      if (!kernel_available(n, dev)) dev = cpu;
    } else {
      // Colocate with a ref-typed input's producer (keeps variables and their
      // updates/readers together); else default to GPU when a kernel exists.
      for (auto* e : n->in_edges) {
        if (e->IsControl()) continue;
        if (e->src_output < (int)e->src->out_is_ref.size() &&
            e->src->out_is_ref[e->src_output] &&
            !e->src->assigned_device.empty()) {
          Device* cand = devices_.LookUp(e->src->assigned_device);
          if (cand && kernel_available(n, cand)) {
            dev = cand;
            break;
          }
        }
      }
      if (!dev) {
        if (gpu0 && kernel_available(n, gpu0)) dev = gpu0;
        else dev = cpu;
      }
    }
    // A node with no kernel anywhere may still be pruned away (dangling
    // gradient subgraphs); defer the hard error to executor creation.
    n->assigned_device = CanonicalDevice(dev->name());
  }

  // ---- 2. Feed rewrite: consumers of fed tensors read a client _Recv. ----
  std::map<std::string, Node*> feed_nodes;  // "name:port" -> recv node
  for (auto& f : feeds) {
    auto [name, port] = ParseTensorName(f);
    Node* n = graph->FindNode(name);
    if (!n) return errors::NotFound("Feed node not found: ", name);
    std::string fkey = name + ":" + std::to_string(port);
    if (feed_nodes.count(fkey)) continue;
    std::string dev = n->assigned_device;
    std::string node_name = "_feed_" + name + "_" + std::to_string(port);
    NodeDef rd = MakeSendRecv("_Recv", node_name, n->out_types[port],
                              "feed:" + fkey, "client", dev);
    Node* recv;
    STF_RETURN_IF_ERROR(graph->AddNode(rd, &recv));
    recv->assigned_device = dev;
    ek->feed_devices["feed:" + fkey] = dev;
    std::vector<Edge*> to_rewire;
    for (auto* e : n->out_edges)
      if (!e->IsControl() && e->src_output == port) to_rewire.push_back(e);
    for (auto* e : to_rewire) {
      Node* dst = e->dst;
      int slot = e->dst_input;
      graph->RemoveEdge(e);
      graph->AddEdge(recv, 0, dst, slot);
    }
    feed_nodes[fkey] = recv;
  }

  // ---- 3. Fetch rewrite: _Send each fetched tensor to the client. ----
  std::vector<Node*> keep;
  for (auto& f : fetches) {
    auto [name, port] = ParseTensorName(f);
    std::string fkey = name + ":" + std::to_string(port);
    Node* src;
    int src_port;
    if (feed_nodes.count(fkey)) {
      src = feed_nodes[fkey];
      src_port = 0;
    } else {
      src = graph->FindNode(name);
      if (!src) return errors::NotFound("Fetch node not found: ", name);
      src_port = port;
      if (src_port >= src->num_outputs())
        return errors::InvalidArgument("Fetch ", f, ": port out of range");
    }
    std::string dev = src->assigned_device;
    NodeDef sd = MakeSendRecv("_Send",
                              "_fetch_" + name + "_" + std::to_string(port) +
                                  "_" + std::to_string(keep.size()),
                              src->out_types[src_port], "fetch:" + fkey, dev,
                              "client");
    Node* send;
    STF_RETURN_IF_ERROR(graph->AddNode(sd, &send));
    send->assigned_device = dev;
    graph->AddEdge(src, src_port, send, 0);
    ek->fetch_devices["fetch:" + fkey] = dev;
    keep.push_back(send);
  }
  for (auto& t : targets) {
    Node* n = graph->FindNode(ParseTensorName(t).first);
    if (!n) return errors::NotFound("Target node not found: ", t);
    keep.push_back(n);
  }

  // ---- 4. Prune to nodes reachable from keep. ----
  std::set<Node*> reachable;
  std::vector<Node*> stack(keep);
  while (!stack.empty()) {
    Node* n = stack.back();
    stack.pop_back();
    if (!reachable.insert(n).second) continue;
    for (auto* e : n->in_edges) stack.push_back(e->src);
  }
  std::vector<Node*> to_remove;
  for (Node* n : graph->nodes())
    if (!reachable.count(n)) to_remove.push_back(n);
  for (Node* n : to_remove) graph->RemoveNode(n);

  // ---- 5. Partition by device, inserting _Send/_Recv pairs. ----
  std::set<std::string> part_devices;
  for (Node* n : graph->nodes()) part_devices.insert(n->assigned_device);

  // Build per-partition GraphDefs.
  std::map<std::string, GraphDef> parts;
  std::map<std::string, std::string> send_dedup;  // src:port|dstdev -> recv name
  int edge_id = 0;
  for (auto& dev : part_devices) parts[dev];  // create empty

  // Emit nodes.
  for (Node* n : graph->nodes()) {
    NodeDef d = n->def;
    d.device = n->assigned_device;
    d.input.clear();
    parts[n->assigned_device].node.push_back(d);
  }
  // Index of nodedef per partition for appending inputs. Indices (not
  // pointers): the node vectors grow while we insert send/recv pairs.
  std::map<std::string, size_t> defs;
  for (auto& kv : parts)
    for (size_t i = 0; i < kv.second.node.size(); ++i)
      defs[kv.first + "|" + kv.second.node[i].name] = i;

  auto add_input = [&](const std::string& dev, const std::string& node,
                       const std::string& input) {
    parts[dev].node[defs[dev + "|" + node]].input.push_back(input);
  };

  for (Node* n : graph->nodes()) {
    // Order data inputs by slot, then controls (NodeDef convention).
    std::vector<const Edge*> data(n->num_inputs(), nullptr);
    std::vector<const Edge*> ctrl;
    for (auto* e : n->in_edges) {
      if (e->IsControl()) ctrl.push_back(e);
      else data[e->dst_input] = e;
    }
    const std::string& ddev = n->assigned_device;
    for (auto* e : data) {
      if (!e) return errors::Internal("missing input edge on ", n->name());
      Node* s = e->src;
      if (s->assigned_device == ddev) {
        add_input(ddev, n->name(),
                  e->src_output == 0
                      ? s->name()
                      : s->name() + ":" + std::to_string(e->src_output));
      } else {
        std::string dkey = s->name() + ":" + std::to_string(e->src_output) +
                           "|" + ddev;
        auto it = send_dedup.find(dkey);
        std::string recv_name;
        if (it != send_dedup.end()) {
          recv_name = it->second;
        } else {
          std::string tname = "e" + std::to_string(edge_id++) + "_" + s->name();
          DataType dt = s->out_types[e->src_output];
          NodeDef sd = MakeSendRecv("_Send", "_s_" + tname, dt, tname,
                                    s->assigned_device, ddev);
          sd.device = s->assigned_device;
          sd.input.push_back(e->src_output == 0
                                 ? s->name()
                                 : s->name() + ":" +
                                       std::to_string(e->src_output));
          parts[s->assigned_device].node.push_back(sd);
          NodeDef rd = MakeSendRecv("_Recv", "_r_" + tname, dt, tname,
                                    s->assigned_device, ddev);
          rd.device = ddev;
          parts[ddev].node.push_back(rd);
          recv_name = rd.name;
          send_dedup[dkey] = recv_name;
        }
        add_input(ddev, n->name(), recv_name);
      }
    }
    for (auto* e : ctrl) {
      Node* s = e->src;
      if (s->assigned_device == ddev) {
        add_input(ddev, n->name(), "^" + s->name());
      } else {
        std::string dkey = s->name() + ":ctrl|" + ddev;
        auto it = send_dedup.find(dkey);
        std::string recv_name;
        if (it != send_dedup.end()) {
          recv_name = it->second;
        } else {
          std::string tname = "c" + std::to_string(edge_id++) + "_" + s->name();
          // Dummy const carrying the control signal.
          NodeDef cd;
          cd.name = "_c_" + tname;
          cd.op = "Const";
          cd.attr["dtype"] = AttrValue::Type(DT_BOOL);
          TensorProto tp;
          tp.dtype = DT_BOOL;
          tp.has_shape = true;
          tp.bool_val.push_back(0);
          AttrValue av;
          av.kind = 'e';
          av.tensor = tp;
          cd.attr["value"] = av;
          cd.device = s->assigned_device;
          cd.input.push_back("^" + s->name());
          parts[s->assigned_device].node.push_back(cd);
          NodeDef sd = MakeSendRecv("_Send", "_s_" + tname, DT_BOOL, tname,
                                    s->assigned_device, ddev);
          sd.device = s->assigned_device;
          sd.input.push_back(cd.name);
          parts[s->assigned_device].node.push_back(sd);
          NodeDef rd = MakeSendRecv("_Recv", "_r_" + tname, DT_BOOL, tname,
                                    s->assigned_device, ddev);
          rd.device = ddev;
          parts[ddev].node.push_back(rd);
          recv_name = rd.name;
          send_dedup[dkey] = recv_name;
        }
        add_input(ddev, n->name(), "^" + recv_name);
      }
    }
  }

  // ---- capture eligibility: one GPU partition; other partitions carry only
  // constants/sends (no per-step host state) ----
  {
    int gpu_parts = 0;
    bool host_trivial = true;
    for (auto& kv : parts) {
      Device* dev = devices_.LookUp(kv.first);
      if (dev && dev->is_gpu()) {
        ++gpu_parts;
        continue;
      }
      for (auto& nd : kv.second.node) {
        if (nd.op != "Const" && nd.op != "_Send" && nd.op != "_Recv" &&
            nd.op != "NoOp" && nd.op != "Shape")
          host_trivial = false;
      }
    }
    ek->capture_eligible = gpu_parts == 1 && host_trivial &&
                           getenv("STF_NO_HIPGRAPH") == nullptr;
  }

  // ---- 6. Build one executor per partition. ----
  for (auto& kv : parts) {
    Device* dev = devices_.LookUp(kv.first);
    if (!dev) return errors::Internal("Unknown partition device ", kv.first);
    auto pg = std::make_unique<Graph>();
    STF_RETURN_IF_ERROR(ConvertGraphDefToGraph(kv.second, pg.get()));
    for (Node* n : pg->nodes()) n->assigned_device = kv.first;
    ExecutorsAndKeys::Item item;
    item.device = dev;
    STF_RETURN_IF_ERROR(
        Executor::Create(std::move(pg), dev, &opseg_, &item.executor));
    ek->items.push_back(std::move(item));
  }
  *out = std::move(ek);
  return Status::OK();
}

void DirectSession::Reset() {
  std::lock_guard<std::mutex> l(mu_);
  executors_.clear();
  opseg_.Clear();
  DeleteResourceMgr(resource_mgr_);
  resource_mgr_ = NewResourceMgr();
}


// ---------------------------------------------------------------------------
// Partial run (reference direct_session.cc PRunSetup/PRun): the executors
// for the full feed/fetch union start immediately; _Recv nodes wait
// asynchronously on the step rendezvous, so feeds can arrive across several
// PartialRun calls and fetches are pulled as their subgraphs complete.
// ---------------------------------------------------------------------------
struct DirectSession::PartialRunState {
  std::shared_ptr<Rendezvous> rendez;
  ExecutorsAndKeys* ek = nullptr;
  std::mutex mu;
  std::condition_variable cv;
  int remaining = 0;
  Status agg;
  std::set<std::string> pending_fetches;  // "name:port" not yet returned
  std::set<std::string> pending_feeds;
};

Status DirectSession::PartialRunSetup(const std::vector<std::string>& feeds,
                                      const std::vector<std::string>& fetches,
                                      const std::vector<std::string>& targets,
                                      std::string* handle) {
  ExecutorsAndKeys* ek = nullptr;
  STF_RETURN_IF_ERROR(GetOrCreateExecutors(feeds, fetches, targets, &ek));
  auto prs = std::make_shared<PartialRunState>();
  prs->rendez = std::make_shared<Rendezvous>();
  prs->ek = ek;
  prs->remaining = (int)ek->items.size();
  for (auto& f : feeds) {
    auto [name, port] = ParseTensorName(f);
    prs->pending_feeds.insert(name + ":" + std::to_string(port));
  }
  for (auto& f : fetches) {
    auto [name, port] = ParseTensorName(f);
    prs->pending_fetches.insert(name + ":" + std::to_string(port));
  }
  int64_t step_id;
  {
    std::lock_guard<std::mutex> l(mu_);
    step_id = ++step_counter_;
    *handle = "prun_" + std::to_string(++partial_run_counter_);
    partial_runs_[*handle] = prs;
  }
  ThreadPool* pool = pool_.get();
  for (auto& item : ek->items) {
    ExecutorArgs args;
    args.step_id = step_id;
    args.rendezvous = prs->rendez.get();
    args.schedule = [pool](std::function<void()> fn) {
      pool->Schedule(std::move(fn));
    };
    args.resource_mgr = resource_mgr_;
    item.executor->RunAsync(args, [prs](Status s) {
      std::lock_guard<std::mutex> l(prs->mu);
      if (!s.ok() && prs->agg.ok()) prs->agg = s;
      if (--prs->remaining == 0) prs->cv.notify_all();
    });
  }
  return Status::OK();
}

Status DirectSession::PartialRun(
    const std::string& handle,
    const std::vector<std::pair<std::string, Tensor>>& feeds,
    const std::vector<std::string>& fetches, std::vector<Tensor>* outputs) {
  std::shared_ptr<PartialRunState> prs;
  {
    std::lock_guard<std::mutex> l(mu_);
    auto it = partial_runs_.find(handle);
    if (it == partial_runs_.end())
      return errors::InvalidArgument("Unknown partial-run handle ", handle);
    prs = it->second;
  }
  for (auto& f : feeds) {
    auto [name, port] = ParseTensorName(f.first);
    std::string fkey = name + ":" + std::to_string(port);
    {
      std::lock_guard<std::mutex> l(prs->mu);
      if (!prs->pending_feeds.erase(fkey))
        return errors::InvalidArgument(
            "Feed ", f.first,
            " was not declared in partial_run_setup (or already fed)");
    }
    auto it = prs->ek->feed_devices.find("feed:" + fkey);
    if (it == prs->ek->feed_devices.end())
      return errors::Internal("feed not wired: ", f.first);
    STF_RETURN_IF_ERROR(prs->rendez->Send(
        RendezvousKey("client", it->second, "feed:" + fkey, "", 0), f.second,
        false));
  }
  outputs->clear();
  Status fetch_status;
  for (auto& f : fetches) {
    auto [name, port] = ParseTensorName(f);
    std::string fkey = name + ":" + std::to_string(port);
    {
      std::lock_guard<std::mutex> l(prs->mu);
      if (!prs->pending_fetches.count(fkey))
        return errors::InvalidArgument(
            "Fetch ", f, " was not declared in partial_run_setup (or was "
            "already fetched)");
    }
    auto it = prs->ek->fetch_devices.find("fetch:" + fkey);
    if (it == prs->ek->fetch_devices.end()) {
      fetch_status = errors::Internal("fetch not wired: ", f);
      break;
    }
    Tensor val;
    bool is_dead = false;
    Status s = prs->rendez->Recv(
        RendezvousKey(it->second, "client", "fetch:" + fkey, "", 0), &val,
        &is_dead);
    if (!s.ok()) {
      fetch_status = s;
      break;
    }
    if (val.IsInitialized() && val.mem_space() == MemSpace::DEVICE) {
      Device* dev = devices_.LookUp(it->second);
      Tensor host;
      Status cs = dev->CopyDeviceTensorToHost(val, &host);
      if (!cs.ok()) {
        fetch_status = cs;
        break;
      }
      val = host;
    }
    outputs->push_back(val);
    std::lock_guard<std::mutex> l(prs->mu);
    prs->pending_fetches.erase(fkey);
  }
  bool finished;
  {
    std::lock_guard<std::mutex> l(prs->mu);
    finished = prs->pending_fetches.empty();
  }
  if (!fetch_status.ok() || finished) {
    // tear down: abort anything still pending, wait for executors
    prs->rendez->StartAbort(errors::Cancelled("partial run finished"));
    {
      std::unique_lock<std::mutex> l(prs->mu);
      prs->cv.wait_for(l, std::chrono::seconds(2),
                       [&]() { return prs->remaining == 0; });
    }
    std::lock_guard<std::mutex> l(mu_);
    partial_runs_.erase(handle);
  }
  return fetch_status;
}

Status DirectSession::Run(
    const std::vector<std::pair<std::string, Tensor>>& feeds,
    const std::vector<std::string>& fetches,
    const std::vector<std::string>& targets, std::vector<Tensor>* outputs,
    StatsCollector* stats, int64_t timeout_ms) {
  std::vector<std::string> feed_names;
  for (auto& f : feeds) feed_names.push_back(f.first);
  ExecutorsAndKeys* ek = nullptr;
  STF_RETURN_IF_ERROR(GetOrCreateExecutors(feed_names, fetches, targets, &ek));

  // Per-step rendezvous on the heap: a timed-out Run returns while stale
  // async work (a cancelled dequeue's continuation) may still complete an
  // executor later; the executor's completion lambda keeps this alive.
  auto rendez_sp = std::make_shared<Rendezvous>();
  Rendezvous& rendez = *rendez_sp;
  // Send feeds.
  for (auto& f : feeds) {
    auto [name, port] = ParseTensorName(f.first);
    std::string fkey = "feed:" + name + ":" + std::to_string(port);
    auto it = ek->feed_devices.find(fkey);
    if (it == ek->feed_devices.end())
      return errors::Internal("feed not wired: ", f.first);
    STF_RETURN_IF_ERROR(rendez.Send(
        RendezvousKey("client", it->second, fkey, "", 0), f.second, false));
  }

  // hipGraph fast path: replay the captured step.
  bool no_io = feeds.empty() && fetches.empty() && stats == nullptr;
  if (no_io && ek->capture_eligible && ek->graph_exec) {
    outputs->clear();
    return ek->capture_device->LaunchCapturedGraph(ek->graph_exec);
  }
  bool do_capture = false;
  Device* capture_dev = nullptr;
  if (no_io && ek->capture_eligible && !ek->capture_broken &&
      ek->graph_exec == nullptr) {
    std::lock_guard<std::mutex> l(mu_);
    if (++ek->plain_runs > 2) {
      do_capture = true;
      for (auto& item : ek->items)
        if (item.device->is_gpu()) capture_dev = item.device;
    }
  }
  if (do_capture && capture_dev) {
    Status s = capture_dev->BeginGraphCapture();
    if (!s.ok()) {
      ek->capture_broken = true;
      do_capture = false;
    }
  }

  // Hardware trace: when stats are requested and a GPU partition exists,
  // bracket every GPU node with stream events (gpu/gpu_tracer.h).
  GpuTracer tracer;
  if (stats) {
    std::vector<Device*> gpus;
    for (auto& item : ek->items)
      if (item.device->is_gpu()) gpus.push_back(item.device);
    if (!gpus.empty() && tracer.Start(gpus).ok()) {
      stats->gpu_pre = [&tracer](Device* d) { return tracer.Pre(d); };
      stats->gpu_post = [&tracer](Device* d, void* t, const std::string& n,
                                  const std::string& o) {
        tracer.Post(d, t, n, o);
      };
    }
  }

  // Run all partition executors.
  int64_t step_id;
  {
    std::lock_guard<std::mutex> l(mu_);
    step_id = ++step_counter_;
  }
  // Completion state lives on the heap: a timed-out Run returns while
  // executor callbacks may still fire (e.g. a dequeue blocked forever), and
  // they must not touch a dead stack frame.
  struct RunState {
    std::mutex mu;
    std::condition_variable cv;
    int remaining = 0;
    Status agg;
    std::atomic<bool> done{false};
    std::atomic<bool> cancelled{false};
  };
  auto rs = std::make_shared<RunState>();
  rs->remaining = (int)ek->items.size();
  std::mutex& mu = rs->mu;
  std::condition_variable& cv = rs->cv;
  int& remaining = rs->remaining;
  Status& agg = rs->agg;
  // Deadline watchdog (RunOptions.timeout_in_ms / ConfigProto
  // operation_timeout_in_ms analog): aborts the step's rendezvous, which
  // unblocks the client fetch Recv and every executor recv.
  std::shared_ptr<std::thread> watchdog;
  if (timeout_ms > 0) {
    watchdog = std::make_shared<std::thread>([rs, rendez_sp, timeout_ms]() {
      std::unique_lock<std::mutex> l(rs->mu);
      if (!rs->cv.wait_for(l, std::chrono::milliseconds(timeout_ms),
                           [&]() { return rs->done.load(); })) {
        rs->cancelled = true;
        l.unlock();
        rendez_sp->StartAbort(
            errors::DeadlineExceeded("Session::Run timed out after ",
                                     timeout_ms, " ms"));
      }
    });
  }
  auto finish_watchdog = [&]() {
    if (watchdog) {
      {
        std::lock_guard<std::mutex> l(rs->mu);
        rs->done = true;
      }
      rs->cv.notify_all();
      watchdog->join();
      watchdog.reset();
    }
  };
  // During hipGraph capture every enqueue must come from this thread: use a
  // single-threaded trampoline queue instead of the pool.
  std::deque<std::function<void()>> inline_q;
  std::mutex inline_mu;
  ThreadPool* pool = pool_.get();
  std::function<void(std::function<void()>)> scheduler;
  if (do_capture) {
    scheduler = [&](std::function<void()> fn) {
      std::lock_guard<std::mutex> l(inline_mu);
      inline_q.push_back(std::move(fn));
    };
  } else {
    scheduler = [pool](std::function<void()> fn) {
      pool->Schedule(std::move(fn));
    };
  }
  for (auto& item : ek->items) {
    ExecutorArgs args;
    args.step_id = step_id;
    args.rendezvous = rendez_sp.get();
    args.is_cancelled = [rs]() { return rs->cancelled.load(); };
    args.schedule = scheduler;
    args.resource_mgr = resource_mgr_;
    args.stats = stats;
    item.executor->RunAsync(args, [rs, rendez_sp](Status s) {
      std::lock_guard<std::mutex> l(rs->mu);
      if (!s.ok() && rs->agg.ok()) rs->agg = s;
      if (--rs->remaining == 0) rs->cv.notify_all();
    });
  }
  if (do_capture) {
    // Drain the trampoline on this thread until all executors complete.
    for (;;) {
      std::function<void()> fn;
      {
        std::lock_guard<std::mutex> l(inline_mu);
        if (!inline_q.empty()) {
          fn = std::move(inline_q.front());
          inline_q.pop_front();
        }
      }
      if (fn) {
        fn();
        continue;
      }
      std::lock_guard<std::mutex> l(mu);
      if (remaining == 0) break;
      // nothing queued but not done: an async kernel callback may still be
      // pending on another thread (not expected in capture-eligible graphs)
      std::this_thread::yield();
    }
  }

  // Collect fetches (they arrive as the graph runs).
  outputs->clear();
  Status fetch_status;
  for (auto& f : fetches) {
    auto [name, port] = ParseTensorName(f);
    std::string fkey = "fetch:" + name + ":" + std::to_string(port);
    auto it = ek->fetch_devices.find(fkey);
    if (it == ek->fetch_devices.end()) {
      fetch_status = errors::Internal("fetch not wired: ", f);
      break;
    }
    Tensor val;
    bool is_dead = false;
    Status s =
        rendez.Recv(RendezvousKey(it->second, "client", fkey, "", 0), &val,
                    &is_dead);
    if (!s.ok()) {
      fetch_status = s;
      break;
    }
    if (val.IsInitialized() && val.mem_space() == MemSpace::DEVICE) {
      Device* dev = devices_.LookUp(it->second);
      Tensor host;
      Status cs = dev->CopyDeviceTensorToHost(val, &host);
      if (!cs.ok()) {
        fetch_status = cs;
        break;
      }
      val = host;
    }
    outputs->push_back(val);
  }

  {
    std::unique_lock<std::mutex> l(mu);
    if (timeout_ms > 0) {
      // Executors unblocked by the abort finish almost immediately; only a
      // kernel stuck outside the rendezvous (queue waiter) stays pending —
      // don't hold the client for it.
      if (!cv.wait_for(l, std::chrono::milliseconds(500),
                       [&]() { return remaining == 0; })) {
        l.unlock();
        finish_watchdog();
        return errors::DeadlineExceeded(
            "Session::Run timed out after ", timeout_ms,
            " ms (executors still pending)");
      }
    } else {
      cv.wait(l, [&]() { return remaining == 0; });
    }
  }
  finish_watchdog();
  if (do_capture && capture_dev) {
    if (agg.ok()) {
      void* exec = nullptr;
      Status cs = capture_dev->EndGraphCapture(&exec);
      if (cs.ok()) {
        ek->graph_exec = exec;
        ek->capture_device = capture_dev;
        // Stream capture records but does not execute: launch the graph once
        // now so this Run() still performs exactly one step.
        Status ls = capture_dev->LaunchCapturedGraph(exec);
        if (!ls.ok()) return ls;
      } else {
        LOG(WARN) << "hipGraph capture failed; falling back to eager: "
                  << cs.ToString();
        ek->capture_broken = true;
      }
    } else {
      void* exec = nullptr;
      capture_dev->EndGraphCapture(&exec);  // abort capture
      ek->capture_broken = true;
    }
  }
  if (stats) {
    stats->gpu_pre = nullptr;
    stats->gpu_post = nullptr;
    (void)tracer.Collect(stats);
  }
  if (!agg.ok()) {
    if (do_capture) {
      // The capture attempt failed mid-step (e.g. a host-dependent op needed
      // a d2h copy). Stream capture executed nothing on the device, so the
      // step can simply be retried eagerly; capture_broken is already set so
      // the retry (and all later steps) skip capture.
      return Run(feeds, fetches, targets, outputs);
    }
    return agg;
  }
  return fetch_status;
}

}  // namespace stf
