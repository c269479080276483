// Gradient-bucket pack/unpack kernels for the fused RCCL all-reduce path
// (rccl/rccl_ops.cc RcclBucketAllReduce).
//
// A bucket is up to STF_COMM_MAX_SEG gradient tensors (mixed f32/bf16)
// flattened into one contiguous f32 staging buffer: pack converts each
// segment to f32, a single ncclAllReduce reduces the flat buffer in f32,
// and unpack scales by 1/world and converts back to each segment's dtype.
// Both kernels run on the dedicated comm stream so they overlap with
// backprop compute on the compute stream.
//
// Segment tables travel by value in the kernarg block (<4 KB on CDNA4) so
// no per-step H2D table upload is needed: tensor addresses change every
// step under BFC reuse.
#include "hip_common.h"

#define STF_COMM_MAX_SEG 120

namespace {

struct SegTable {
  const void* ptr[STF_COMM_MAX_SEG];  // per-segment tensor base
  int64_t end[STF_COMM_MAX_SEG];      // exclusive flat end offset (elements)
  unsigned char is_bf16[STF_COMM_MAX_SEG];
  int nseg;
};

__device__ __forceinline__ int FindSeg(const SegTable& t, int64_t idx) {
  // first segment whose end > idx (ends are sorted ascending)
  int lo = 0, hi = t.nseg - 1;
  while (lo < hi) {
    int mid = (lo + hi) >> 1;
    if (t.end[mid] > idx) hi = mid; else lo = mid + 1;
  }
  return lo;
}

__global__ void CommPackKernel(SegTable t, float* __restrict__ dst,
                               int64_t total) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int s = FindSeg(t, i);
    int64_t base = s == 0 ? 0 : t.end[s - 1];
    int64_t off = i - base;
    if (t.is_bf16[s]) {
      const uint16_t* src = (const uint16_t*)t.ptr[s];
      dst[i] = bf16_to_f32(src[off]);
    } else {
      const float* src = (const float*)t.ptr[s];
      dst[i] = src[off];
    }
  }
}

// ptr[] entries are the OUTPUT tensors here (non-const; cast inside).
__global__ void CommUnpackKernel(SegTable t, const float* __restrict__ src,
                                 float scale, int64_t total) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    int s = FindSeg(t, i);
    int64_t base = s == 0 ? 0 : t.end[s - 1];
    int64_t off = i - base;
    float v = src[i] * scale;
    if (t.is_bf16[s]) {
      uint16_t* dst = (uint16_t*)t.ptr[s];
      dst[off] = f32_to_bf16(v);
    } else {
      float* dst = (float*)t.ptr[s];
      dst[off] = v;
    }
  }
}

}  // namespace

extern "C" {

int stf_comm_max_segments() { return STF_COMM_MAX_SEG; }

hipError_t stf_comm_pack(int nseg, const void* const* ptrs,
                         const int64_t* ends, const unsigned char* is_bf16,
                         float* dst, int64_t total, hipStream_t stream) {
  if (nseg <= 0 || nseg > STF_COMM_MAX_SEG) return hipErrorInvalidValue;
  SegTable t;
  t.nseg = nseg;
  for (int i = 0; i < nseg; ++i) {
    t.ptr[i] = ptrs[i];
    t.end[i] = ends[i];
    t.is_bf16[i] = is_bf16[i];
  }
  hipLaunchKernelGGL(CommPackKernel, ElemwiseGrid(total, 256, 4), dim3(256),
                     0, stream, t, dst, total);
  return hipGetLastError();
}

hipError_t stf_comm_unpack(int nseg, void* const* ptrs, const int64_t* ends,
                           const unsigned char* is_bf16, const float* src,
                           float scale, int64_t total, hipStream_t stream) {
  if (nseg <= 0 || nseg > STF_COMM_MAX_SEG) return hipErrorInvalidValue;
  SegTable t;
  t.nseg = nseg;
  for (int i = 0; i < nseg; ++i) {
    t.ptr[i] = ptrs[i];
    t.end[i] = ends[i];
    t.is_bf16[i] = is_bf16[i];
  }
  hipLaunchKernelGGL(CommUnpackKernel, ElemwiseGrid(total, 256, 4), dim3(256),
                     0, stream, t, src, scale, total);
  return hipGetLastError();
}

}  // extern "C"
