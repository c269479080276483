// Hand-defined message structs wire-compatible with the reference's
// core/framework protos (graph.proto, node_def.proto, attr_value.proto,
// tensor.proto, tensor_shape.proto, versions.proto). Field numbers match the
// reference schemas so serialized GraphDefs/TensorProtos interoperate.
#pragma once

#include <map>
#include <string>
#include <vector>

#include "core/base.h"
#include "core/pb.h"

namespace stf {

// tensor_shape.proto: TensorShapeProto { Dim dim = 2 {size=1,name=2};
//                                        unknown_rank = 3 }
struct TensorShapeProto {
  struct Dim {
    int64_t size = 0;
    std::string name;
  };
  std::vector<Dim> dim;
  bool unknown_rank = false;

  void Serialize(pb::Writer* w) const;
  bool Parse(pb::Reader* r);
  TensorShape AsShape() const {
    TensorShape s;
    for (auto& d : dim) s.AddDim(d.size);
    return s;
  }
  static TensorShapeProto From(const TensorShape& s) {
    TensorShapeProto p;
    for (auto d : s.dim_sizes()) {
      Dim dd;
      dd.size = d;
      p.dim.push_back(dd);
    }
    return p;
  }
};

// tensor.proto: TensorProto { dtype=1, tensor_shape=2, version_number=3,
//   tensor_content=4, float_val=5, double_val=6, int_val=7, string_val=8,
//   int64_val=10, bool_val=11, half_val=13 }
struct TensorProto {
  DataType dtype = DT_INVALID;
  TensorShapeProto tensor_shape;
  bool has_shape = false;
  std::string tensor_content;
  std::vector<float> float_val;
  std::vector<double> double_val;
  std::vector<int32_t> int_val;
  std::vector<std::string> string_val;
  std::vector<int64_t> int64_val;
  std::vector<int32_t> bool_val;
  std::vector<int32_t> half_val;  // also bfloat16 values

  void Serialize(pb::Writer* w) const;
  bool Parse(pb::Reader* r);
};

struct AttrValue;

// attr_value.proto: ListValue { s=2,i=3,f=4,b=5,type=6,shape=7,tensor=8 }
struct AttrListValue {
  std::vector<std::string> s;
  std::vector<int64_t> i;
  std::vector<float> f;
  std::vector<bool> b;
  std::vector<int> type;  // DataType
  std::vector<TensorShapeProto> shape;
  std::vector<TensorProto> tensor;

  void Serialize(pb::Writer* w) const;
  bool Parse(pb::Reader* r);
};

// attr_value.proto: AttrValue { list=1,s=2,i=3,f=4,b=5,type=6,shape=7,
//   tensor=8,placeholder=9,func=10 }
struct AttrValue {
  // which oneof member is set: 0=unset,'s','i','f','b','t'(type),'h'(shape),
  // 'e'(tensor),'l'(list),'p'(placeholder)
  char kind = 0;
  std::string s;
  int64_t i = 0;
  float f = 0;
  bool b = false;
  DataType type = DT_INVALID;
  TensorShapeProto shape;
  TensorProto tensor;
  AttrListValue list;
  std::string placeholder;

  void Serialize(pb::Writer* w) const;
  bool Parse(pb::Reader* r);

  static AttrValue S(const std::string& v) { AttrValue a; a.kind = 's'; a.s = v; return a; }
  static AttrValue I(int64_t v) { AttrValue a; a.kind = 'i'; a.i = v; return a; }
  static AttrValue F(float v) { AttrValue a; a.kind = 'f'; a.f = v; return a; }
  static AttrValue B(bool v) { AttrValue a; a.kind = 'b'; a.b = v; return a; }
  static AttrValue Type(DataType v) { AttrValue a; a.kind = 't'; a.type = v; return a; }
  static AttrValue Shape(const TensorShapeProto& v) { AttrValue a; a.kind = 'h'; a.shape = v; return a; }
};

// node_def.proto: NodeDef { name=1, op=2, input=3, device=4, attr=5(map) }
struct NodeDef {
  std::string name;
  std::string op;
  std::vector<std::string> input;
  std::string device;
  std::map<std::string, AttrValue> attr;

  void Serialize(pb::Writer* w) const;
  bool Parse(pb::Reader* r);
  std::string DebugString() const { return op + "(" + name + ")"; }
};

// versions.proto: VersionDef { producer=1, min_consumer=2, bad_consumers=3 }
struct VersionDef {
  int32_t producer = 0;
  int32_t min_consumer = 0;
  void Serialize(pb::Writer* w) const;
  bool Parse(pb::Reader* r);
};

// graph.proto: GraphDef { node=1, library=2, version=3(deprecated),
//                         versions=4 }
struct GraphDef {
  std::vector<NodeDef> node;
  VersionDef versions;

  std::string SerializeAsString() const;
  bool ParseFromString(const std::string& data);
};

// Helpers used across the runtime.
inline bool GetAttrInt(const NodeDef& n, const std::string& key, int64_t* out) {
  auto it = n.attr.find(key);
  if (it == n.attr.end() || it->second.kind != 'i') return false;
  *out = it->second.i;
  return true;
}
inline bool GetAttrType(const NodeDef& n, const std::string& key, DataType* out) {
  auto it = n.attr.find(key);
  if (it == n.attr.end() || it->second.kind != 't') return false;
  *out = it->second.type;
  return true;
}
inline bool GetAttrString(const NodeDef& n, const std::string& key, std::string* out) {
  auto it = n.attr.find(key);
  if (it == n.attr.end() || it->second.kind != 's') return false;
  *out = it->second.s;
  return true;
}
inline bool GetAttrBool(const NodeDef& n, const std::string& key, bool* out) {
  auto it = n.attr.find(key);
  if (it == n.attr.end() || it->second.kind != 'b') return false;
  *out = it->second.b;
  return true;
}

}  // namespace stf
