#include "core/protos.h"

namespace stf {

// --------------------------- TensorShapeProto ------------------------------
void TensorShapeProto::Serialize(pb::Writer* w) const {
  for (auto& d : dim) {
    pb::Writer sub;
    sub.PutInt64(1, d.size);
    sub.PutString(2, d.name);
    w->PutMessage(2, sub.buf());
  }
  w->PutBool(3, unknown_rank);
}

bool TensorShapeProto::Parse(pb::Reader* r) {
  int field, wire;
  while (r->ReadTag(&field, &wire)) {
    if (field == 2 && wire == 2) {
      const char* d;
      size_t l;
      if (!r->ReadView(&d, &l)) return false;
      pb::Reader sub(d, l);
      Dim dd;
      int f2, w2;
      while (sub.ReadTag(&f2, &w2)) {
        if (f2 == 1 && w2 == 0) {
          uint64_t v;
          if (!sub.ReadVarint(&v)) return false;
          dd.size = (int64_t)v;
        } else if (f2 == 2 && w2 == 2) {
          if (!sub.ReadBytes(&dd.name)) return false;
        } else if (!sub.SkipField(w2)) {
          return false;
        }
      }
      dim.push_back(dd);
    } else if (field == 3 && wire == 0) {
      uint64_t v;
      if (!r->ReadVarint(&v)) return false;
      unknown_rank = v != 0;
    } else if (!r->SkipField(wire)) {
      return false;
    }
  }
  return true;
}

// ------------------------------ TensorProto --------------------------------
void TensorProto::Serialize(pb::Writer* w) const {
  w->PutInt64(1, (int64_t)dtype);
  if (has_shape || !tensor_shape.dim.empty()) {
    pb::Writer sub;
    tensor_shape.Serialize(&sub);
    w->PutMessage(2, sub.buf());
  }
  w->PutString(4, tensor_content);
  w->PutPackedFloats(5, float_val);
  if (!double_val.empty()) {
    pb::Writer sub;
    for (double v : double_val) {
      char tmp[8];
      std::memcpy(tmp, &v, 8);
      sub.buf().append(tmp, 8);
    }
    w->PutMessage(6, sub.buf());
  }
  if (!int_val.empty()) {
    std::vector<int64_t> tmp(int_val.begin(), int_val.end());
    w->PutPackedVarints(7, tmp);
  }
  for (auto& s : string_val) w->PutStringAlways(8, s);
  w->PutPackedVarints(10, int64_val);
  if (!bool_val.empty()) {
    std::vector<int64_t> tmp(bool_val.begin(), bool_val.end());
    w->PutPackedVarints(11, tmp);
  }
  if (!half_val.empty()) {
    std::vector<int64_t> tmp(half_val.begin(), half_val.end());
    w->PutPackedVarints(13, tmp);
  }
}

bool TensorProto::Parse(pb::Reader* r) {
  int field, wire;
  while (r->ReadTag(&field, &wire)) {
    switch (field) {
      case 1: {
        uint64_t v;
        if (!r->ReadVarint(&v)) return false;
        dtype = (DataType)v;
        break;
      }
      case 2: {
        const char* d;
        size_t l;
        if (!r->ReadView(&d, &l)) return false;
        pb::Reader sub(d, l);
        if (!tensor_shape.Parse(&sub)) return false;
        has_shape = true;
        break;
      }
      case 4:
        if (!r->ReadBytes(&tensor_content)) return false;
        break;
      case 5: {
        if (wire == 2) {
          const char* d;
          size_t l;
          if (!r->ReadView(&d, &l)) return false;
          for (size_t i = 0; i + 4 <= l; i += 4) {
            float f;
            std::memcpy(&f, d + i, 4);
            float_val.push_back(f);
          }
        } else {
          uint32_t v;
          if (!r->ReadFixed32(&v)) return false;
          float f;
          std::memcpy(&f, &v, 4);
          float_val.push_back(f);
        }
        break;
      }
      case 6: {
        if (wire == 2) {
          const char* d;
          size_t l;
          if (!r->ReadView(&d, &l)) return false;
          for (size_t i = 0; i + 8 <= l; i += 8) {
            double f;
            std::memcpy(&f, d + i, 8);
            double_val.push_back(f);
          }
        } else {
          uint64_t v;
          if (!r->ReadFixed64(&v)) return false;
          double f;
          std::memcpy(&f, &v, 8);
          double_val.push_back(f);
        }
        break;
      }
      case 7:
      case 10:
      case 11:
      case 13: {
        std::vector<int64_t> vals;
        if (wire == 2) {
          const char* d;
          size_t l;
          if (!r->ReadView(&d, &l)) return false;
          pb::Reader sub(d, l);
          uint64_t v;
          while (!sub.done()) {
            if (!sub.ReadVarint(&v)) return false;
            vals.push_back((int64_t)v);
          }
        } else {
          uint64_t v;
          if (!r->ReadVarint(&v)) return false;
          vals.push_back((int64_t)v);
        }
        for (int64_t v : vals) {
          if (field == 7) int_val.push_back((int32_t)v);
          else if (field == 10) int64_val.push_back(v);
          else if (field == 11) bool_val.push_back((int32_t)v);
          else half_val.push_back((int32_t)v);
        }
        break;
      }
      case 8: {
        std::string s;
        if (!r->ReadBytes(&s)) return false;
        string_val.push_back(std::move(s));
        break;
      }
      default:
        if (!r->SkipField(wire)) return false;
    }
  }
  return true;
}

// ----------------------------- AttrListValue -------------------------------
void AttrListValue::Serialize(pb::Writer* w) const {
  for (auto& v : s) w->PutStringAlways(2, v);
  w->PutPackedVarints(3, i);
  w->PutPackedFloats(4, f);
  if (!b.empty()) {
    std::vector<int64_t> tmp;
    for (bool v : b) tmp.push_back(v ? 1 : 0);
    w->PutPackedVarints(5, tmp);
  }
  if (!type.empty()) {
    std::vector<int64_t> tmp(type.begin(), type.end());
    w->PutPackedVarints(6, tmp);
  }
  for (auto& v : shape) {
    pb::Writer sub;
    v.Serialize(&sub);
    w->PutMessage(7, sub.buf());
  }
  for (auto& v : tensor) {
    pb::Writer sub;
    v.Serialize(&sub);
    w->PutMessage(8, sub.buf());
  }
}

bool AttrListValue::Parse(pb::Reader* r) {
  int field, wire;
  while (r->ReadTag(&field, &wire)) {
    switch (field) {
      case 2: {
        std::string v;
        if (!r->ReadBytes(&v)) return false;
        s.push_back(std::move(v));
        break;
      }
      case 3:
      case 5:
      case 6: {
        std::vector<int64_t> vals;
        if (wire == 2) {
          const char* d;
          size_t l;
          if (!r->ReadView(&d, &l)) return false;
          pb::Reader sub(d, l);
          uint64_t v;
          while (!sub.done()) {
            if (!sub.ReadVarint(&v)) return false;
            vals.push_back((int64_t)v);
          }
        } else {
          uint64_t v;
          if (!r->ReadVarint(&v)) return false;
          vals.push_back((int64_t)v);
        }
        for (int64_t v : vals) {
          if (field == 3) i.push_back(v);
          else if (field == 5) b.push_back(v != 0);
          else type.push_back((int)v);
        }
        break;
      }
      case 4: {
        if (wire == 2) {
          const char* d;
          size_t l;
          if (!r->ReadView(&d, &l)) return false;
          for (size_t k = 0; k + 4 <= l; k += 4) {
            float fv;
            std::memcpy(&fv, d + k, 4);
            f.push_back(fv);
          }
        } else {
          uint32_t v;
          if (!r->ReadFixed32(&v)) return false;
          float fv;
          std::memcpy(&fv, &v, 4);
          f.push_back(fv);
        }
        break;
      }
      case 7: {
        const char* d;
        size_t l;
        if (!r->ReadView(&d, &l)) return false;
        pb::Reader sub(d, l);
        TensorShapeProto p;
        if (!p.Parse(&sub)) return false;
        shape.push_back(std::move(p));
        break;
      }
      case 8: {
        const char* d;
        size_t l;
        if (!r->ReadView(&d, &l)) return false;
        pb::Reader sub(d, l);
        TensorProto p;
        if (!p.Parse(&sub)) return false;
        tensor.push_back(std::move(p));
        break;
      }
      default:
        if (!r->SkipField(wire)) return false;
    }
  }
  return true;
}

// ------------------------------- AttrValue ---------------------------------
void AttrValue::Serialize(pb::Writer* w) const {
  switch (kind) {
    case 'l': {
      pb::Writer sub;
      list.Serialize(&sub);
      w->PutMessage(1, sub.buf());
      break;
    }
    case 's':
      w->PutStringAlways(2, s);
      break;
    case 'i':
      w->PutTag(3, 0);
      w->PutVarint((uint64_t)i);
      break;
    case 'f': {
      w->PutTag(4, 5);
      char tmp[4];
      std::memcpy(tmp, &f, 4);
      w->buf().append(tmp, 4);
      break;
    }
    case 'b':
      w->PutTag(5, 0);
      w->PutVarint(b ? 1 : 0);
      break;
    case 't':
      w->PutTag(6, 0);
      w->PutVarint((uint64_t)type);
      break;
    case 'h': {
      pb::Writer sub;
      shape.Serialize(&sub);
      w->PutMessage(7, sub.buf());
      break;
    }
    case 'e': {
      pb::Writer sub;
      tensor.Serialize(&sub);
      w->PutMessage(8, sub.buf());
      break;
    }
    case 'p':
      w->PutStringAlways(9, placeholder);
      break;
    default:
      break;
  }
}

bool AttrValue::Parse(pb::Reader* r) {
  int field, wire;
  while (r->ReadTag(&field, &wire)) {
    switch (field) {
      case 1: {
        const char* d;
        size_t l;
        if (!r->ReadView(&d, &l)) return false;
        pb::Reader sub(d, l);
        if (!list.Parse(&sub)) return false;
        kind = 'l';
        break;
      }
      case 2:
        if (!r->ReadBytes(&s)) return false;
        kind = 's';
        break;
      case 3: {
        uint64_t v;
        if (!r->ReadVarint(&v)) return false;
        i = (int64_t)v;
        kind = 'i';
        break;
      }
      case 4: {
        uint32_t v;
        if (!r->ReadFixed32(&v)) return false;
        std::memcpy(&f, &v, 4);
        kind = 'f';
        break;
      }
      case 5: {
        uint64_t v;
        if (!r->ReadVarint(&v)) return false;
        b = v != 0;
        kind = 'b';
        break;
      }
      case 6: {
        uint64_t v;
        if (!r->ReadVarint(&v)) return false;
        type = (DataType)v;
        kind = 't';
        break;
      }
      case 7: {
        const char* d;
        size_t l;
        if (!r->ReadView(&d, &l)) return false;
        pb::Reader sub(d, l);
        if (!shape.Parse(&sub)) return false;
        kind = 'h';
        break;
      }
      case 8: {
        const char* d;
        size_t l;
        if (!r->ReadView(&d, &l)) return false;
        pb::Reader sub(d, l);
        if (!tensor.Parse(&sub)) return false;
        kind = 'e';
        break;
      }
      case 9:
        if (!r->ReadBytes(&placeholder)) return false;
        kind = 'p';
        break;
      default:
        if (!r->SkipField(wire)) return false;
    }
  }
  return true;
}

// -------------------------------- NodeDef ----------------------------------
void NodeDef::Serialize(pb::Writer* w) const {
  w->PutString(1, name);
  w->PutString(2, op);
  for (auto& in : input) w->PutStringAlways(3, in);
  w->PutString(4, device);
  for (auto& kv : attr) {
    // map<string, AttrValue> entry: key=1, value=2
    pb::Writer entry;
    entry.PutString(1, kv.first);
    pb::Writer val;
    kv.second.Serialize(&val);
    entry.PutMessage(2, val.buf());
    w->PutMessage(5, entry.buf());
  }
}

bool NodeDef::Parse(pb::Reader* r) {
  int field, wire;
  while (r->ReadTag(&field, &wire)) {
    switch (field) {
      case 1:
        if (!r->ReadBytes(&name)) return false;
        break;
      case 2:
        if (!r->ReadBytes(&op)) return false;
        break;
      case 3: {
        std::string v;
        if (!r->ReadBytes(&v)) return false;
        input.push_back(std::move(v));
        break;
      }
      case 4:
        if (!r->ReadBytes(&device)) return false;
        break;
      case 5: {
        const char* d;
        size_t l;
        if (!r->ReadView(&d, &l)) return false;
        pb::Reader sub(d, l);
        std::string key;
        AttrValue val;
        int f2, w2;
        while (sub.ReadTag(&f2, &w2)) {
          if (f2 == 1 && w2 == 2) {
            if (!sub.ReadBytes(&key)) return false;
          } else if (f2 == 2 && w2 == 2) {
            const char* vd;
            size_t vl;
            if (!sub.ReadView(&vd, &vl)) return false;
            pb::Reader vr(vd, vl);
            if (!val.Parse(&vr)) return false;
          } else if (!sub.SkipField(w2)) {
            return false;
          }
        }
        attr[key] = std::move(val);
        break;
      }
      default:
        if (!r->SkipField(wire)) return false;
    }
  }
  return true;
}

// ------------------------------- VersionDef --------------------------------
void VersionDef::Serialize(pb::Writer* w) const {
  w->PutInt64(1, producer);
  w->PutInt64(2, min_consumer);
}

bool VersionDef::Parse(pb::Reader* r) {
  int field, wire;
  while (r->ReadTag(&field, &wire)) {
    if (field == 1 && wire == 0) {
      uint64_t v;
      if (!r->ReadVarint(&v)) return false;
      producer = (int32_t)v;
    } else if (field == 2 && wire == 0) {
      uint64_t v;
      if (!r->ReadVarint(&v)) return false;
      min_consumer = (int32_t)v;
    } else if (!r->SkipField(wire)) {
      return false;
    }
  }
  return true;
}

// -------------------------------- GraphDef ---------------------------------
std::string GraphDef::SerializeAsString() const {
  pb::Writer w;
  for (auto& n : node) {
    pb::Writer sub;
    n.Serialize(&sub);
    w.PutMessage(1, sub.buf());
  }
  pb::Writer v;
  versions.Serialize(&v);
  if (!v.buf().empty()) w.PutMessage(4, v.buf());
  return w.buf();
}

bool GraphDef::ParseFromString(const std::string& data) {
  pb::Reader r(data);
  int field, wire;
  while (r.ReadTag(&field, &wire)) {
    if (field == 1 && wire == 2) {
      const char* d;
      size_t l;
      if (!r.ReadView(&d, &l)) return false;
      pb::Reader sub(d, l);
      NodeDef n;
      if (!n.Parse(&sub)) return false;
      node.push_back(std::move(n));
    } else if (field == 4 && wire == 2) {
      const char* d;
      size_t l;
      if (!r.ReadView(&d, &l)) return false;
      pb::Reader sub(d, l);
      if (!versions.Parse(&sub)) return false;
    } else if (!r.SkipField(wire)) {
      return false;
    }
  }
  return true;
}

}  // namespace stf
