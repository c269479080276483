// Op registry: OpDef + REGISTER_OP with a spec-string builder, mirroring the
// capability of the reference's OpRegistry/OpDefBuilder
// (reference: tensorflow/core/framework/op.h, op_def_builder.cc) with a
// compact parser. Shape inference lives in the Python layer.
#pragma once

#include <map>
#include <set>
#include <string>
#include <vector>

#include "core/base.h"
#include "core/protos.h"

namespace stf {

struct OpDef {
  struct ArgDef {
    std::string name;
    DataType type = DT_INVALID;   // concrete type, if fixed
    std::string type_attr;        // attr holding the type ("T")
    std::string number_attr;      // attr holding repeat count ("N")
    std::string type_list_attr;   // attr holding a list of types
    bool is_ref = false;
  };
  struct AttrDef {
    std::string name;
    std::string type;  // "int","float","bool","string","type","shape",
                       // "list(int)","list(float)","list(string)","list(type)","list(shape)"
    bool has_default = false;
    AttrValue default_value;
    std::vector<DataType> allowed;  // for "type" attrs with {..} constraint
    bool has_minimum = false;
    int64_t minimum = 0;
  };

  std::string name;
  std::vector<ArgDef> input_arg;
  std::vector<ArgDef> output_arg;
  std::vector<AttrDef> attr;
  bool is_stateful = false;
  bool allows_uninitialized_input = false;

  const AttrDef* FindAttr(const std::string& n) const {
    for (auto& a : attr)
      if (a.name == n) return &a;
    return nullptr;
  }
};

class OpDefBuilder {
 public:
  explicit OpDefBuilder(const std::string& name) { def_.name = name; }
  OpDefBuilder& Input(const std::string& spec);
  OpDefBuilder& Output(const std::string& spec);
  OpDefBuilder& Attr(const std::string& spec);
  OpDefBuilder& SetIsStateful() {
    def_.is_stateful = true;
    return *this;
  }
  OpDefBuilder& SetAllowsUninitializedInput() {
    def_.allows_uninitialized_input = true;
    return *this;
  }
  const OpDef& Build() const { return def_; }

 private:
  OpDef def_;
};

class OpRegistry {
 public:
  static OpRegistry* Global();
  void Register(const OpDef& def);
  const OpDef* LookUp(const std::string& op) const;
  std::vector<std::string> ListOps() const;

 private:
  std::map<std::string, OpDef> ops_;
};

namespace register_op {
struct OpDefBuilderReceiver {
  OpDefBuilderReceiver(const OpDefBuilder& b) {  // NOLINT
    OpRegistry::Global()->Register(b.Build());
  }
};
}  // namespace register_op

#define REGISTER_OP_UNIQ_HELPER(ctr, name) \
  static ::stf::register_op::OpDefBuilderReceiver register_op##ctr = \
      ::stf::OpDefBuilder(name)
#define REGISTER_OP_UNIQ(ctr, name) REGISTER_OP_UNIQ_HELPER(ctr, name)
#define REGISTER_OP(name) REGISTER_OP_UNIQ(__COUNTER__, name)

// ---- NodeDef type resolution -----------------------------------------------
// Resolve the full input/output DataType lists of `node` against its OpDef,
// expanding number_attr/type_list_attr. Ref-ness is dropped (all types
// returned as base types); is_ref flags returned separately when requested.
Status InOutTypesForNode(const NodeDef& node, const OpDef& op_def,
                         std::vector<DataType>* in_types,
                         std::vector<DataType>* out_types,
                         std::vector<bool>* out_is_ref = nullptr);

// Fill in defaults for attrs not present in node.attr.
void AddDefaultsToNodeDef(const OpDef& op_def, NodeDef* node);

}  // namespace stf
