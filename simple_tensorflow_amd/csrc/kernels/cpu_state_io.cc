// SaveV2 / RestoreV2 / MergeV2Checkpoints kernels over the tensor-bundle
// format (capability + format analog of the reference's
// kernels/save_restore_v2_ops.cc:90,141,186 + util/tensor_bundle: data shards
// `prefix.data-NNNNN-of-MMMMM` with raw tensor bytes, `prefix.index` as a
// LevelDB-format table of BundleEntryProto values keyed by tensor name, ""
// key holding BundleHeaderProto).
#include <fstream>
#include <sys/stat.h>

#include "core/pb.h"
#include "kernels/kernel_util.h"
#include "util/table.h"

namespace stf {

namespace {

std::string ShardName(const std::string& prefix, int shard, int num_shards) {
  char buf[64];
  snprintf(buf, sizeof(buf), ".data-%05d-of-%05d", shard, num_shards);
  return prefix + buf;
}

Status WriteFileString(const std::string& path, const std::string& data) {
  // ensure parent dir exists (single level best-effort)
  auto slash = path.rfind('/');
  if (slash != std::string::npos) {
    std::string dir = path.substr(0, slash);
    mkdir(dir.c_str(), 0755);
  }
  std::ofstream f(path, std::ios::binary | std::ios::trunc);
  if (!f) return errors::Internal("cannot open ", path, " for write");
  f.write(data.data(), (std::streamsize)data.size());
  f.close();
  if (!f) return errors::Internal("write failed: ", path);
  return Status::OK();
}

Status ReadFileString(const std::string& path, std::string* out) {
  std::ifstream f(path, std::ios::binary);
  if (!f) return errors::NotFound("file not found: ", path);
  f.seekg(0, std::ios::end);
  out->resize((size_t)f.tellg());
  f.seekg(0);
  f.read(&(*out)[0], (std::streamsize)out->size());
  if (!f) return errors::Internal("read failed: ", path);
  return Status::OK();
}

// BundleEntryProto: dtype=1, shape=2, shard_id=3, offset=4, size=5,
// crc32c=6(fixed32)
std::string EncodeBundleEntry(DataType dt, const TensorShape& shape,
                              int shard, int64_t offset, int64_t size,
                              uint32_t crc) {
  pb::Writer w;
  w.PutInt64(1, (int64_t)dt);
  pb::Writer sh;
  TensorShapeProto::From(shape).Serialize(&sh);
  w.PutMessage(2, sh.buf());
  w.PutInt64(3, shard);
  w.PutInt64(4, offset);
  w.PutInt64(5, size);
  w.PutTag(6, 5);
  char tmp[4];
  std::memcpy(tmp, &crc, 4);
  w.buf().append(tmp, 4);
  return w.buf();
}

struct BundleEntry {
  DataType dtype = DT_INVALID;
  TensorShape shape;
  int shard = 0;
  int64_t offset = 0, size = 0;
  uint32_t crc = 0;
};

Status DecodeBundleEntry(const std::string& data, BundleEntry* e) {
  pb::Reader r(data);
  int field, wire;
  while (r.ReadTag(&field, &wire)) {
    switch (field) {
      case 1: {
        uint64_t v;
        if (!r.ReadVarint(&v)) return errors::InvalidArgument("entry dtype");
        e->dtype = (DataType)v;
        break;
      }
      case 2: {
        const char* d;
        size_t l;
        if (!r.ReadView(&d, &l)) return errors::InvalidArgument("entry shape");
        pb::Reader sub(d, l);
        TensorShapeProto p;
        if (!p.Parse(&sub)) return errors::InvalidArgument("entry shape");
        e->shape = p.AsShape();
        break;
      }
      case 3: {
        uint64_t v;
        r.ReadVarint(&v);
        e->shard = (int)v;
        break;
      }
      case 4: {
        uint64_t v;
        r.ReadVarint(&v);
        e->offset = (int64_t)v;
        break;
      }
      case 5: {
        uint64_t v;
        r.ReadVarint(&v);
        e->size = (int64_t)v;
        break;
      }
      case 6: {
        uint32_t v;
        if (!r.ReadFixed32(&v)) return errors::InvalidArgument("entry crc");
        e->crc = v;
        break;
      }
      default:
        if (!r.SkipField(wire)) return errors::InvalidArgument("entry");
    }
  }
  return Status::OK();
}

}  // namespace

// --------------------------------- SaveV2 ----------------------------------
class SaveV2Op : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const std::string& prefix = ctx->input(0).flat<std::string>()[0];
    const Tensor& names = ctx->input(1);
    const Tensor& slices = ctx->input(2);
    int n = (int)names.NumElements();
    OP_REQUIRES(ctx, num_inputs() == n + 3,
                errors::InvalidArgument("SaveV2: tensor count mismatch"));
    std::string data;
    std::map<std::string, std::string> index;
    // sorted by name for the table; remember per-tensor order first
    std::vector<int> order(n);
    for (int i = 0; i < n; ++i) order[i] = i;
    std::sort(order.begin(), order.end(), [&](int a, int b) {
      return names.flat<std::string>()[a] < names.flat<std::string>()[b];
    });
    for (int oi : order) {
      const Tensor& t = ctx->input(3 + oi);
      OP_REQUIRES(ctx, slices.flat<std::string>()[oi].empty(),
                  errors::Unimplemented("SaveV2 slices"));
      // Raw-byte copy would serialize heap pointers for DT_STRING tensors;
      // reject until a length-prefixed string encoding exists
      // (reference tensor_bundle.cc serializes strings specially).
      OP_REQUIRES(ctx, t.dtype() != DT_STRING,
                  errors::Unimplemented("SaveV2 does not support DT_STRING ",
                                        "tensors (tensor ",
                                        names.flat<std::string>()[oi], ")"));
      int64_t offset = (int64_t)data.size();
      int64_t size = (int64_t)t.TotalBytes();
      data.append((const char*)t.raw_data(), size);
      uint32_t crc = table::MaskCrc(
          table::Crc32c((const char*)t.raw_data(), size));
      index[names.flat<std::string>()[oi]] = EncodeBundleEntry(
          t.dtype(), t.shape(), 0, offset, size, crc);
    }
    // header at key "": BundleHeaderProto {num_shards=1, endianness=0,
    // version={producer=1}}
    pb::Writer h;
    h.PutInt64(1, 1);
    pb::Writer v;
    v.PutInt64(1, 1);
    h.PutMessage(3, v.buf());
    index[""] = h.buf();
    std::string table_bytes;
    OP_REQUIRES_OK(ctx, table::BuildTable(index, &table_bytes));
    OP_REQUIRES_OK(ctx, WriteFileString(ShardName(prefix, 0, 1), data));
    OP_REQUIRES_OK(ctx, WriteFileString(prefix + ".index", table_bytes));
  }
};
REGISTER_KERNEL_BUILDER(Name("SaveV2").Device(DEVICE_CPU), SaveV2Op);

// -------------------------------- RestoreV2 ---------------------------------
class RestoreV2Op : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const std::string& prefix = ctx->input(0).flat<std::string>()[0];
    const Tensor& names = ctx->input(1);
    int n = (int)names.NumElements();
    std::string table_bytes;
    OP_REQUIRES_OK(ctx, ReadFileString(prefix + ".index", &table_bytes));
    std::map<std::string, std::string> index;
    OP_REQUIRES_OK(ctx, table::ReadTable(table_bytes, &index));
    // shard count from header
    int num_shards = 1;
    auto hit = index.find("");
    if (hit != index.end()) {
      pb::Reader r(hit->second);
      int field, wire;
      while (r.ReadTag(&field, &wire)) {
        if (field == 1 && wire == 0) {
          uint64_t v;
          r.ReadVarint(&v);
          num_shards = (int)v;
        } else if (!r.SkipField(wire)) {
          break;
        }
      }
    }
    std::map<int, std::string> shard_data;
    for (int i = 0; i < n; ++i) {
      const std::string& name = names.flat<std::string>()[i];
      auto it = index.find(name);
      OP_REQUIRES(ctx, it != index.end(),
                  errors::NotFound("tensor ", name, " not in checkpoint ",
                                   prefix));
      BundleEntry e;
      OP_REQUIRES_OK(ctx, DecodeBundleEntry(it->second, &e));
      auto& data = shard_data[e.shard];
      if (data.empty())
        OP_REQUIRES_OK(
            ctx, ReadFileString(ShardName(prefix, e.shard, num_shards), &data));
      OP_REQUIRES(ctx, e.offset + e.size <= (int64_t)data.size(),
                  errors::InvalidArgument("bundle entry out of range"));
      // Verify the stored masked crc32c before accepting the bytes so a
      // corrupted checkpoint fails loudly (reference tensor_bundle.cc:580).
      uint32_t got = table::MaskCrc(
          table::Crc32c(data.data() + e.offset, (size_t)e.size));
      OP_REQUIRES(ctx, e.crc == 0 || got == e.crc,
                  errors::DataLoss("checksum mismatch restoring ", name,
                                   " from ", prefix));
      Tensor* out = ctx->allocate_output(i, e.shape);
      OP_REQUIRES(ctx, out->dtype() == e.dtype,
                  errors::InvalidArgument("restore dtype mismatch for ", name));
      std::memcpy(out->raw_data(), data.data() + e.offset, e.size);
    }
  }
};
REGISTER_KERNEL_BUILDER(Name("RestoreV2").Device(DEVICE_CPU), RestoreV2Op);

// ---------------------------- MergeV2Checkpoints ----------------------------
// Round 1: single-shard saves; merge just renames metadata (parity stub that
// handles the MonitoredSession call pattern).
class MergeV2CheckpointsOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& src = ctx->input(0);
    const std::string& dst = ctx->input(1).flat<std::string>()[0];
    const std::string& sp = src.flat<std::string>()[0];
    std::string idx;
    OP_REQUIRES_OK(ctx, ReadFileString(sp + ".index", &idx));
    OP_REQUIRES_OK(ctx, WriteFileString(dst + ".index", idx));
    std::string data;
    OP_REQUIRES_OK(ctx, ReadFileString(ShardName(sp, 0, 1), &data));
    OP_REQUIRES_OK(ctx, WriteFileString(ShardName(dst, 0, 1), data));
  }
};
REGISTER_KERNEL_BUILDER(Name("MergeV2Checkpoints").Device(DEVICE_CPU),
                        MergeV2CheckpointsOp);

}  // namespace stf
