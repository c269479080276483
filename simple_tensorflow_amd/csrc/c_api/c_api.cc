// C API implementation over DirectSession (capability analog of reference
// c/c_api.cc; see c_api.h for the entry points mirrored).
#include "c_api/c_api.h"

#include <cstring>
#include <fstream>
#include <map>
#include <memory>
#include <string>
#include <vector>

#include "core/pb.h"
#include "core/protos.h"
#include "core/tensor.h"
#include "runtime/session.h"

using namespace stf;

struct TF_Status {
  Status s;
};
struct TF_Tensor {
  Tensor t;
};
struct TF_SessionOptions {};
struct TF_ImportGraphDefOptions {
  std::string prefix;
};
struct TF_Operation {
  std::string name;
  std::string op;
};
struct TF_Graph {
  GraphDef gdef;
  std::map<std::string, std::unique_ptr<TF_Operation>> opers;
  TF_Operation* Find(const std::string& name) {
    auto it = opers.find(name);
    if (it != opers.end()) return it->second.get();
    for (auto& n : gdef.node) {
      if (n.name == name) {
        auto op = std::make_unique<TF_Operation>();
        op->name = n.name;
        op->op = n.op;
        TF_Operation* raw = op.get();
        opers[name] = std::move(op);
        return raw;
      }
    }
    return nullptr;
  }
};
struct TF_Session {
  DirectSession sess;
  TF_Graph* graph = nullptr;
  explicit TF_Session(bool cpu_only = false) : sess(cpu_only) {}
};

extern "C" {

TF_Status* TF_NewStatus(void) { return new TF_Status(); }
void TF_DeleteStatus(TF_Status* s) { delete s; }
void TF_SetStatus(TF_Status* s, TF_Code code, const char* msg) {
  s->s = Status((stf::Code)code, msg ? msg : "");
}
TF_Code TF_GetCode(const TF_Status* s) { return (TF_Code)s->s.code(); }
const char* TF_Message(const TF_Status* s) {
  static thread_local std::string msg;
  msg = s->s.message();
  return msg.c_str();
}

static void BufferFreeData(void* data, size_t) { free(data); }

TF_Buffer* TF_NewBuffer(void) {
  TF_Buffer* b = new TF_Buffer();
  b->data = nullptr;
  b->length = 0;
  b->data_deallocator = nullptr;
  return b;
}
TF_Buffer* TF_NewBufferFromString(const void* proto, size_t len) {
  TF_Buffer* b = TF_NewBuffer();
  void* copy = malloc(len);
  memcpy(copy, proto, len);
  b->data = copy;
  b->length = len;
  b->data_deallocator = BufferFreeData;
  return b;
}
void TF_DeleteBuffer(TF_Buffer* b) {
  if (!b) return;
  if (b->data_deallocator && b->data)
    b->data_deallocator(const_cast<void*>(b->data), b->length);
  delete b;
}

static void SetBuffer(TF_Buffer* b, const std::string& s) {
  if (b->data_deallocator && b->data)
    b->data_deallocator(const_cast<void*>(b->data), b->length);
  void* copy = malloc(s.size());
  memcpy(copy, s.data(), s.size());
  b->data = copy;
  b->length = s.size();
  b->data_deallocator = BufferFreeData;
}

TF_Tensor* TF_AllocateTensor(TF_DataType dt, const int64_t* dims,
                             int num_dims, size_t len) {
  TensorShape shape;
  for (int i = 0; i < num_dims; ++i) shape.AddDim(dims[i]);
  TF_Tensor* t = new TF_Tensor{Tensor((DataType)dt, shape)};
  (void)len;
  return t;
}
TF_Tensor* TF_NewTensor(TF_DataType dt, const int64_t* dims, int num_dims,
                        void* data, size_t len,
                        void (*deallocator)(void*, size_t, void*),
                        void* deallocator_arg) {
  TF_Tensor* t = TF_AllocateTensor(dt, dims, num_dims, len);
  memcpy(t->t.raw_data(), data, std::min(len, t->t.TotalBytes()));
  if (deallocator) deallocator(data, len, deallocator_arg);
  return t;
}
void TF_DeleteTensor(TF_Tensor* t) { delete t; }
TF_DataType TF_TensorType(const TF_Tensor* t) {
  return (TF_DataType)t->t.dtype();
}
int TF_NumDims(const TF_Tensor* t) { return t->t.dims(); }
int64_t TF_Dim(const TF_Tensor* t, int i) { return t->t.dim_size(i); }
size_t TF_TensorByteSize(const TF_Tensor* t) { return t->t.TotalBytes(); }
void* TF_TensorData(const TF_Tensor* t) {
  return const_cast<TF_Tensor*>(t)->t.raw_data();
}

TF_Graph* TF_NewGraph(void) { return new TF_Graph(); }
void TF_DeleteGraph(TF_Graph* g) { delete g; }
TF_ImportGraphDefOptions* TF_NewImportGraphDefOptions(void) {
  return new TF_ImportGraphDefOptions();
}
void TF_DeleteImportGraphDefOptions(TF_ImportGraphDefOptions* o) { delete o; }
void TF_ImportGraphDefOptionsSetPrefix(TF_ImportGraphDefOptions* o,
                                       const char* prefix) {
  o->prefix = prefix ? prefix : "";
}

void TF_GraphImportGraphDef(TF_Graph* graph, const TF_Buffer* graph_def,
                            const TF_ImportGraphDefOptions* options,
                            TF_Status* status) {
  GraphDef gd;
  std::string data((const char*)graph_def->data, graph_def->length);
  if (!gd.ParseFromString(data)) {
    status->s = errors::InvalidArgument("Invalid GraphDef");
    return;
  }
  std::string prefix =
      options && !options->prefix.empty() ? options->prefix + "/" : "";
  for (auto& n : gd.node) {
    NodeDef copy = n;
    if (!prefix.empty()) {
      copy.name = prefix + copy.name;
      for (auto& in : copy.input) {
        if (!in.empty() && in[0] == '^')
          in = "^" + prefix + in.substr(1);
        else
          in = prefix + in;
      }
    }
    graph->gdef.node.push_back(std::move(copy));
  }
  status->s = Status::OK();
}

void TF_GraphToGraphDef(TF_Graph* graph, TF_Buffer* out, TF_Status* status) {
  SetBuffer(out, graph->gdef.SerializeAsString());
  status->s = Status::OK();
}

TF_Operation* TF_GraphOperationByName(TF_Graph* graph, const char* name) {
  return graph->Find(name);
}
const char* TF_OperationName(TF_Operation* oper) { return oper->name.c_str(); }
const char* TF_OperationOpType(TF_Operation* oper) { return oper->op.c_str(); }

TF_SessionOptions* TF_NewSessionOptions(void) {
  return new TF_SessionOptions();
}
void TF_DeleteSessionOptions(TF_SessionOptions* o) { delete o; }

TF_Session* TF_NewSession(TF_Graph* graph, const TF_SessionOptions*,
                          TF_Status* status) {
  auto* s = new TF_Session();
  s->graph = graph;
  status->s = s->sess.Create(graph->gdef);
  if (!status->s.ok()) {
    delete s;
    return nullptr;
  }
  return s;
}
void TF_CloseSession(TF_Session*, TF_Status* status) {
  status->s = Status::OK();
}
void TF_DeleteSession(TF_Session* s, TF_Status* status) {
  delete s;
  status->s = Status::OK();
}

void TF_SessionRun(TF_Session* session, const TF_Buffer*,
                   const TF_Output* inputs, TF_Tensor* const* input_values,
                   int ninputs, const TF_Output* outputs,
                   TF_Tensor** output_values, int noutputs,
                   const TF_Operation* const* target_opers, int ntargets,
                   TF_Buffer*, TF_Status* status) {
  std::vector<std::pair<std::string, Tensor>> feeds;
  for (int i = 0; i < ninputs; ++i) {
    std::string name = inputs[i].oper->name;
    if (inputs[i].index > 0) name += ":" + std::to_string(inputs[i].index);
    feeds.emplace_back(name, input_values[i]->t);
  }
  std::vector<std::string> fetches;
  for (int i = 0; i < noutputs; ++i) {
    std::string name = outputs[i].oper->name;
    if (outputs[i].index > 0) name += ":" + std::to_string(outputs[i].index);
    fetches.push_back(name);
  }
  std::vector<std::string> targets;
  for (int i = 0; i < ntargets; ++i) targets.push_back(target_opers[i]->name);
  std::vector<Tensor> results;
  status->s = session->sess.Run(feeds, fetches, targets, &results);
  if (!status->s.ok()) return;
  for (int i = 0; i < noutputs && i < (int)results.size(); ++i)
    output_values[i] = new TF_Tensor{results[i]};
}

// ---- SavedModel loading (c_api.h TF_LoadSessionFromSavedModel) ----
namespace {

// Extracts (graph_def, saver filename tensor, saver restore op) of the meta
// graph whose MetaInfoDef.tags == tags. Field numbers per meta_graph.proto.
bool FindMetaGraph(const std::string& saved_model,
                   const std::vector<std::string>& tags, std::string* mgd_out,
                   std::string* gd, std::string* filename_tensor,
                   std::string* restore_op) {
  pb::Reader top(saved_model);
  int field, wire;
  while (top.ReadTag(&field, &wire)) {
    if (field != 2) {
      top.SkipField(wire);
      continue;
    }
    std::string mgd;
    if (!top.ReadBytes(&mgd)) return false;
    std::vector<std::string> mtags;
    std::string g, fn, ro;
    pb::Reader r(mgd);
    int f2, w2;
    while (r.ReadTag(&f2, &w2)) {
      if (f2 == 1) {
        std::string mi;
        r.ReadBytes(&mi);
        pb::Reader ri(mi);
        int f3, w3;
        while (ri.ReadTag(&f3, &w3)) {
          if (f3 == 4) {
            std::string t;
            ri.ReadBytes(&t);
            mtags.push_back(t);
          } else {
            ri.SkipField(w3);
          }
        }
      } else if (f2 == 2) {
        r.ReadBytes(&g);
      } else if (f2 == 3) {
        std::string sd;
        r.ReadBytes(&sd);
        pb::Reader rs(sd);
        int f3, w3;
        while (rs.ReadTag(&f3, &w3)) {
          if (f3 == 1)
            rs.ReadBytes(&fn);
          else if (f3 == 3)
            rs.ReadBytes(&ro);
          else
            rs.SkipField(w3);
        }
      } else {
        r.SkipField(w2);
      }
    }
    if (mtags.size() == tags.size()) {
      bool all = true;
      for (auto& t : tags) {
        bool found = false;
        for (auto& mt : mtags)
          if (mt == t) found = true;
        all = all && found;
      }
      if (all) {
        *mgd_out = mgd;
        *gd = g;
        *filename_tensor = fn;
        *restore_op = ro;
        return true;
      }
    }
  }
  return false;
}

}  // namespace

TF_Session* TF_LoadSessionFromSavedModel(
    const TF_SessionOptions*, const TF_Buffer*, const char* export_dir,
    const char* const* tags, int tags_len, TF_Graph* graph,
    TF_Buffer* meta_graph_def, TF_Status* status) {
  std::string path = std::string(export_dir) + "/saved_model.pb";
  std::ifstream f(path, std::ios::binary);
  if (!f) {
    status->s = errors::NotFound("SavedModel not found at ", path);
    return nullptr;
  }
  std::string data((std::istreambuf_iterator<char>(f)),
                   std::istreambuf_iterator<char>());
  std::vector<std::string> want;
  for (int i = 0; i < tags_len; ++i) want.push_back(tags[i]);
  std::string mgd, gd_bytes, filename_tensor, restore_op;
  if (!FindMetaGraph(data, want, &mgd, &gd_bytes, &filename_tensor,
                     &restore_op)) {
    status->s = errors::NotFound("No meta graph with requested tags");
    return nullptr;
  }
  if (meta_graph_def) SetBuffer(meta_graph_def, mgd);
  GraphDef gdef;
  if (!gdef.ParseFromString(gd_bytes)) {
    status->s = errors::InvalidArgument("Invalid GraphDef in SavedModel");
    return nullptr;
  }
  graph->gdef = gdef;
  auto* s = new TF_Session();
  s->graph = graph;
  status->s = s->sess.Create(graph->gdef);
  if (!status->s.ok()) {
    delete s;
    return nullptr;
  }
  if (!restore_op.empty() && !filename_tensor.empty()) {
    Tensor fn(DT_STRING, TensorShape({}));
    fn.flat<std::string>()[0] =
        std::string(export_dir) + "/variables/variables";
    std::vector<Tensor> unused;
    auto colon = filename_tensor.rfind(':');
    std::string feed_name = colon == std::string::npos
                                ? filename_tensor
                                : filename_tensor.substr(0, colon);
    auto rc = restore_op.rfind(':');
    std::string target =
        rc == std::string::npos ? restore_op : restore_op.substr(0, rc);
    status->s = s->sess.Run({{feed_name, fn}}, {}, {target}, &unused);
    if (!status->s.ok()) {
      delete s;
      return nullptr;
    }
  }
  return s;
}

const char* TF_Version(void) { return "1.0.1-mi355x"; }

}  // extern "C"
