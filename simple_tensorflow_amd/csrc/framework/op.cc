#include "framework/op.h"

#include <mutex>

namespace stf {

namespace {

// Parse a type expression inside an arg spec: "float", "T", "N * T",
// "N * float", "Ref(...)".
void ParseArgSpec(const std::string& spec, OpDef::ArgDef* arg) {
  auto colon = spec.find(':');
  CHECK(colon != std::string::npos) << "bad arg spec: " << spec;
  arg->name = StrStrip(spec.substr(0, colon));
  std::string type = StrStrip(spec.substr(colon + 1));
  if (StrStartsWith(type, "Ref(")) {
    arg->is_ref = true;
    type = StrStrip(type.substr(4, type.size() - 5));
  }
  auto star = type.find('*');
  if (star != std::string::npos) {
    arg->number_attr = StrStrip(type.substr(0, star));
    type = StrStrip(type.substr(star + 1));
  }
  DataType dt = DataTypeFromString(type);
  if (dt != DT_INVALID) {
    arg->type = dt;
  } else if (StrStartsWith(type, "list(")) {
    // "list(T)" style handled via type_list_attr name inside parens
    arg->type_list_attr = StrStrip(type.substr(5, type.size() - 6));
  } else {
    arg->type_attr = type;
  }
}

AttrValue ParseDefault(const std::string& type, const std::string& text) {
  AttrValue v;
  std::string t = StrStrip(text);
  if (type == "int") {
    v = AttrValue::I(strtoll(t.c_str(), nullptr, 10));
  } else if (type == "float") {
    v = AttrValue::F(strtof(t.c_str(), nullptr));
  } else if (type == "bool") {
    v = AttrValue::B(t == "true");
  } else if (type == "string") {
    if (t.size() >= 2 && (t[0] == '"' || t[0] == '\''))
      t = t.substr(1, t.size() - 2);
    v = AttrValue::S(t);
  } else if (type == "type") {
    DataType dt = DataTypeFromString(t);
    if (dt == DT_INVALID && StrStartsWith(t, "DT_")) {
      std::string lower;
      for (size_t i = 3; i < t.size(); ++i) lower += (char)tolower(t[i]);
      dt = DataTypeFromString(lower);
    }
    v = AttrValue::Type(dt);
  } else if (StrStartsWith(type, "list(")) {
    v.kind = 'l';
    std::string inner = type.substr(5, type.size() - 6);
    if (t.size() >= 2 && t[0] == '[') t = t.substr(1, t.size() - 2);
    if (!StrStrip(t).empty()) {
      for (auto& item : StrSplit(t, ',')) {
        std::string it = StrStrip(item);
        if (inner == "int") v.list.i.push_back(strtoll(it.c_str(), nullptr, 10));
        else if (inner == "float") v.list.f.push_back(strtof(it.c_str(), nullptr));
        else if (inner == "string") {
          if (it.size() >= 2 && (it[0] == '"' || it[0] == '\''))
            it = it.substr(1, it.size() - 2);
          v.list.s.push_back(it);
        }
      }
    }
  } else if (type == "shape") {
    v.kind = 'h';
  }
  return v;
}

}  // namespace

OpDefBuilder& OpDefBuilder::Input(const std::string& spec) {
  OpDef::ArgDef arg;
  ParseArgSpec(spec, &arg);
  def_.input_arg.push_back(arg);
  return *this;
}

OpDefBuilder& OpDefBuilder::Output(const std::string& spec) {
  OpDef::ArgDef arg;
  ParseArgSpec(spec, &arg);
  def_.output_arg.push_back(arg);
  return *this;
}

// Attr spec grammar: "name: type" | "name: type = default" |
// "name: {float, double}" (type attr with allowed set) |
// "name: int >= 1".
OpDefBuilder& OpDefBuilder::Attr(const std::string& spec) {
  OpDef::AttrDef a;
  auto colon = spec.find(':');
  CHECK(colon != std::string::npos) << "bad attr spec: " << spec;
  a.name = StrStrip(spec.substr(0, colon));
  std::string rest = StrStrip(spec.substr(colon + 1));

  // default value
  std::string def_text;
  // Careful: '=' may appear in ">=". Find " = " outside of ">=".
  for (size_t i = 0; i + 1 < rest.size(); ++i) {
    if (rest[i] == '=' && (i == 0 || rest[i - 1] != '>') ) {
      def_text = StrStrip(rest.substr(i + 1));
      rest = StrStrip(rest.substr(0, i));
      break;
    }
  }
  // minimum constraint
  auto ge = rest.find(">=");
  if (ge != std::string::npos) {
    a.has_minimum = true;
    a.minimum = strtoll(rest.substr(ge + 2).c_str(), nullptr, 10);
    rest = StrStrip(rest.substr(0, ge));
  }

  if (!rest.empty() && rest[0] == '{') {
    // allowed-type set, e.g. "{float, double, bfloat16}"
    a.type = "type";
    std::string inner = rest.substr(1, rest.size() - 2);
    for (auto& item : StrSplit(inner, ',')) {
      DataType dt = DataTypeFromString(StrStrip(item));
      if (dt != DT_INVALID) a.allowed.push_back(dt);
    }
  } else if (rest == "numbertype" || rest == "realnumbertype") {
    a.type = "type";
    a.allowed = {DT_FLOAT, DT_DOUBLE, DT_INT32, DT_INT64, DT_BFLOAT16,
                 DT_HALF, DT_UINT8, DT_INT8, DT_INT16, DT_UINT16};
  } else {
    a.type = rest;
  }
  if (!def_text.empty()) {
    a.has_default = true;
    a.default_value = ParseDefault(a.type, def_text);
  }
  def_.attr.push_back(a);
  return *this;
}

OpRegistry* OpRegistry::Global() {
  static OpRegistry* r = new OpRegistry();
  return r;
}

static std::mutex& registry_mu() {
  static std::mutex mu;
  return mu;
}

void OpRegistry::Register(const OpDef& def) {
  std::lock_guard<std::mutex> l(registry_mu());
  OpDef fixed = def;
  // "arg: A" where A is a list(type) attr means a variadic input whose
  // element types come from that list — reclassify type_attr as
  // type_list_attr now that all attrs are known.
  auto fixup = [&](std::vector<OpDef::ArgDef>* args) {
    for (auto& a : *args) {
      if (!a.type_attr.empty()) {
        const OpDef::AttrDef* ad = fixed.FindAttr(a.type_attr);
        if (ad && ad->type == "list(type)") {
          a.type_list_attr = a.type_attr;
          a.type_attr.clear();
        }
      }
    }
  };
  fixup(&fixed.input_arg);
  fixup(&fixed.output_arg);
  ops_[def.name] = fixed;
}

const OpDef* OpRegistry::LookUp(const std::string& op) const {
  std::lock_guard<std::mutex> l(registry_mu());
  auto it = ops_.find(op);
  return it == ops_.end() ? nullptr : &it->second;
}

std::vector<std::string> OpRegistry::ListOps() const {
  std::lock_guard<std::mutex> l(registry_mu());
  std::vector<std::string> out;
  for (auto& kv : ops_) out.push_back(kv.first);
  return out;
}

static Status ResolveArg(const OpDef::ArgDef& arg, const NodeDef& node,
                         const OpDef& op_def, std::vector<DataType>* types,
                         std::vector<bool>* is_ref) {
  int64_t n = 1;
  if (!arg.number_attr.empty()) {
    if (!GetAttrInt(node, arg.number_attr, &n))
      return errors::InvalidArgument("node ", node.name, ": missing attr ",
                                     arg.number_attr);
  }
  if (!arg.type_list_attr.empty()) {
    auto it = node.attr.find(arg.type_list_attr);
    if (it == node.attr.end() || it->second.kind != 'l')
      return errors::InvalidArgument("node ", node.name, ": missing list attr ",
                                     arg.type_list_attr);
    for (int t : it->second.list.type) {
      types->push_back((DataType)t);
      if (is_ref) is_ref->push_back(arg.is_ref);
    }
    return Status::OK();
  }
  DataType dt = arg.type;
  if (!arg.type_attr.empty()) {
    if (!GetAttrType(node, arg.type_attr, &dt))
      return errors::InvalidArgument("node ", node.name, " (", node.op,
                                     "): missing type attr ", arg.type_attr);
  }
  for (int64_t i = 0; i < n; ++i) {
    types->push_back(dt);
    if (is_ref) is_ref->push_back(arg.is_ref);
  }
  return Status::OK();
}

Status InOutTypesForNode(const NodeDef& node, const OpDef& op_def,
                         std::vector<DataType>* in_types,
                         std::vector<DataType>* out_types,
                         std::vector<bool>* out_is_ref) {
  if (in_types)
    for (auto& arg : op_def.input_arg)
      STF_RETURN_IF_ERROR(ResolveArg(arg, node, op_def, in_types, nullptr));
  if (out_types)
    for (auto& arg : op_def.output_arg)
      STF_RETURN_IF_ERROR(ResolveArg(arg, node, op_def, out_types, out_is_ref));
  return Status::OK();
}

void AddDefaultsToNodeDef(const OpDef& op_def, NodeDef* node) {
  for (auto& a : op_def.attr) {
    if (a.has_default && node->attr.find(a.name) == node->attr.end())
      node->attr[a.name] = a.default_value;
  }
}

}  // namespace stf
