// 256-row-tile 8-phase bf16 MFMA GEMM for gfx950 (NT: C[M,N] = A[M,K]*B[N,K]^T).
//
// The top rung of the CDNA4 guide's GEMM ladder (cdna_hip_programming.md §5
// "The 256² 8-phase template"): 8 waves (2M x 4N, 512 threads), LDS ring of
// 16 KiB half-tiles, one C-quadrant of MFMAs per phase with fine
// ds_read ∥ global_load_lds ∥ MFMA interleave, counted vmcnt only at K-tile
// boundaries (never a vmcnt(0) drain in the main loop), s_setprio around the
// MFMA cluster, raw s_barrier (never __syncthreads, whose fence would drain
// the in-flight LDS-DMA).
//
// Generalized over BN (column tile) for the conv-shaped GEMMs:
//   NR=4: 256x256 (128 KiB LDS, 1 block/CU)  — fat matmuls
//   NR=2: 256x128 ( 96 KiB LDS, 1 block/CU)  — Cout/Cin = 128-sized convs
//   NR=1: 256x64  ( 80 KiB LDS, 2 blocks/CU) — stage-1 convs (N=64, 576)
//
// Per-wave: 128 x (NR*16) output = 8(m) x NR 16x16 fragments. B fragments
// are register-cached for the whole K-tile (read once, phase 1); each phase
// reads one A quadrant (2 m-frags x 2 k-steps = 4 ds_read_b128) and runs
// 2 x NR x 2 v_mfma_f32_16x16x32_bf16.
//
// Prefetch schedule (per tile u, phases 1..4):
//   ph1: glds A-half0(u+1)   [slot (u+1)&1: its previous tile u-1 was fully
//   ph2: glds A-half1(u+1)    consumed at u-1's phase 4 gate]
//   ph3: glds B-piece0(u+2)  [slot u&1: B(u) was register-cached at ph1]
//   ph4: glds B-piece1(u+2) (BN=256 only); s_waitcnt vmcnt(B loads of u+2)
//        — everything tile u+1 needs has landed, B(u+2) stays in flight.
#include "hip_common.h"

namespace {

__device__ __forceinline__ f32x4 mfma_bf16_8(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// 16B-chunk swizzle by row (same both-sides rule as gemm_bf16.hip): 16
// consecutive rows at one chunk land on 16 distinct bank positions.
__device__ __forceinline__ int Swz8(int r) { return (r ^ (r >> 3)) & 7; }

constexpr int kBM = 256, kBK = 64;
constexpr int kThreads = 512;

// Stage `ROWS` rows x 64 k with lane-linear glds (ROWS/64 passes of
// 512 x 16B). The source column carries the inverse swizzle so the LDS
// image is the swizzled one. AG maps (row, k) -> global pointer (plain
// matrix or implicit-conv NHWC).
template <int ROWS, class AG>
__device__ __forceinline__ void Stage8(const AG& ag, int64_t row0, int64_t k0,
                                       uint16_t* lds_base, int tid) {
#pragma unroll
  for (int p = 0; p < ROWS / 64; ++p) {
    int s = p * kThreads + tid;          // 16B slot
    int r = s >> 3;
    int c8 = (s & 7) ^ Swz8(r);
    const uint16_t* g = ag.at(row0 + r, k0 + c8 * 8);
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) uint32_t*)g,
        (__attribute__((address_space(3))) uint32_t*)(lds_base + s * 8), 16,
        0, 0);
  }
}

// NR = B fragments per wave; BN = 64*NR (wave grid fixed 2M x 4N).
template <int NR, bool OUT_BF16, bool FUSE_RELU, class AG>
__launch_bounds__(512) __global__ void Gemm256x8Ph(
    AG ag, const uint16_t* __restrict__ B, void* __restrict__ C,
    const float* __restrict__ bias, const uint16_t* __restrict__ side,
    int64_t M, int64_t N, int64_t K, int64_t ldb) {
  constexpr int kBN = 64 * NR;
  constexpr int kABytes = kBM * kBK * 2;       // 32 KiB per slot
  constexpr int kBBytes = kBN * kBK * 2;       // per slot
  extern __shared__ __attribute__((aligned(16))) uint16_t lds[];
  auto a_slot = [&](int slot) { return lds + (size_t)slot * (kABytes / 2); };
  auto b_slot = [&](int slot) {
    return lds + kABytes + (size_t)slot * (kBBytes / 2);
  };

  int nbm = (int)(M / kBM), nbn = (int)(N / kBN);
  int bid = XcdSwizzle(blockIdx.x, nbm * nbn);
  int64_t m0 = (int64_t)(bid / nbn) * kBM;
  int64_t n0 = (int64_t)(bid % nbn) * kBN;

  int tid = threadIdx.x;
  int lane = tid & 63;
  int wid = tid >> 6;          // 0..7
  int wr = wid >> 2;           // 0..1  (A half = wr)
  int wc = wid & 3;            // 0..3

  int nkt = (int)(K / kBK);

  // ---- prologue: A(0), B(0) -> slot 0; B(1) -> slot 1 ----
  Stage8<128>(ag, m0, 0, a_slot(0), tid);
  Stage8<128>(ag, m0 + 128, 0, a_slot(0) + 128 * kBK, tid);
  Stage8<kBN>(LinearAG{B, ldb}, n0, 0, b_slot(0), tid);
  if (nkt > 1) {
    Stage8<kBN>(LinearAG{B, ldb}, n0, kBK, b_slot(1), tid);
    if (NR == 4) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else if (NR == 2) asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    else asm volatile("s_waitcnt vmcnt(1)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
  __builtin_amdgcn_s_barrier();

  f32x4 acc[8][NR];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < NR; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // A: this wave reads rows [wr*128, wr*128+128) of the current slot;
  //    local row = i*16 + (lane&15), chunk = (kk*4 + (lane>>4)) ^ Swz8(row).
  // B: rows [wc*16*NR, +16*NR).
  int a_row_base = wr * 128 + (lane & 15);
  int b_row_base = wc * 16 * NR + (lane & 15);
  int kq = lane >> 4;  // 16B chunk quadrant

  for (int u = 0; u < nkt; ++u) {
    int slot = u & 1;
    const uint16_t* at = a_slot(slot);
    const uint16_t* bt = b_slot(slot);

    bf16x8 bfr[NR][2];
    bf16x8 afr[2][2];

    // ---------------- phase 1: B frags + A quadrant 0 ----------------
#pragma unroll
    for (int j = 0; j < NR; ++j) {
      int r = b_row_base + j * 16;
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        bfr[j][kk] =
            *(const bf16x8*)(bt + (r * 8 + ((kk * 4 + kq) ^ Swz8(r))) * 8);
    }
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int r = a_row_base + i * 16;
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
        afr[i][kk] =
            *(const bf16x8*)(at + (r * 8 + ((kk * 4 + kq) ^ Swz8(r))) * 8);
    }
    if (u + 1 < nkt)
      Stage8<128>(ag, m0, (int64_t)(u + 1) * kBK, a_slot(slot ^ 1), tid);
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int j = 0; j < NR; ++j)
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          acc[i][j] = mfma_bf16_8(afr[i][kk], bfr[j][kk], acc[i][j]);
    __builtin_amdgcn_s_setprio(0);
    __builtin_amdgcn_s_barrier();

    // ---------------- phases 2..4: A quadrants 1..3 ----------------
#pragma unroll
    for (int q = 1; q < 4; ++q) {
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        int r = a_row_base + (q * 2 + i) * 16;
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
          afr[i][kk] =
              *(const bf16x8*)(at + (r * 8 + ((kk * 4 + kq) ^ Swz8(r))) * 8);
      }
      if (q == 1) {
        if (u + 1 < nkt)
          Stage8<128>(ag, m0 + 128, (int64_t)(u + 1) * kBK,
                      a_slot(slot ^ 1) + 128 * kBK, tid);
      } else if (q == 2) {
        if (NR == 4) {
          if (u + 2 < nkt)
            Stage8<128>(LinearAG{B, ldb}, n0, (int64_t)(u + 2) * kBK,
                        b_slot(slot), tid);
        }
      } else {
        if (u + 2 < nkt) {
          if (NR == 4) {
            Stage8<128>(LinearAG{B, ldb}, n0 + 128, (int64_t)(u + 2) * kBK,
                        b_slot(slot) + 128 * kBK, tid);
            asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
          } else if (NR == 2) {
            Stage8<kBN>(LinearAG{B, ldb}, n0, (int64_t)(u + 2) * kBK,
                        b_slot(slot), tid);
            asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
          } else {
            Stage8<kBN>(LinearAG{B, ldb}, n0, (int64_t)(u + 2) * kBK,
                        b_slot(slot), tid);
            asm volatile("s_waitcnt vmcnt(1)" ::: "memory");
          }
        } else {
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
      }
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < NR; ++j)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            acc[q * 2 + i][j] =
                mfma_bf16_8(afr[i][kk], bfr[j][kk], acc[q * 2 + i][j]);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
  }

  // ---- epilogue ----
  if (OUT_BF16) {
    // Stage the bf16 tile through LDS (A/B buffers are dead) and emit
    // contiguous 16B stores.
    uint16_t* cbuf = lds;
#pragma unroll
    for (int i = 0; i < 8; ++i) {
#pragma unroll
      for (int j = 0; j < NR; ++j) {
        int col = wc * 16 * NR + j * 16 + (lane & 15);
        float bv = bias ? bias[n0 + col] : 0.f;
#pragma unroll
        for (int rgi = 0; rgi < 4; ++rgi) {
          int row = wr * 128 + i * 16 + (lane >> 4) * 4 + rgi;
          float v = acc[i][j][rgi] + bv;
          if (FUSE_RELU) v = v > 0.f ? v : 0.f;
          cbuf[row * kBN + col] = f32_to_bf16(v);
        }
      }
    }
    __builtin_amdgcn_s_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
#pragma unroll
    for (int p = 0; p < kBM * kBN / 8 / kThreads; ++p) {
      int sidx = (p * kThreads + tid) * 8;
      int r = sidx / kBN;
      int c = sidx % kBN;
      int64_t off = (m0 + r) * N + n0 + c;
      if (side) {
        // fused elementwise side add (residual-gradient accumulation):
        // one extra 16B load replaces a whole separate add pass
        __bf16 v[8], sv[8];
        *(ulong2*)v = *(ulong2*)(cbuf + sidx);
        *(ulong2*)sv = *(const ulong2*)(side + off);
#pragma unroll
        for (int e = 0; e < 8; ++e) v[e] = (__bf16)((float)v[e] + (float)sv[e]);
        *(ulong2*)((uint16_t*)C + off) = *(ulong2*)v;
      } else {
        *(ulong2*)((uint16_t*)C + off) = *(ulong2*)(cbuf + sidx);
      }
    }
    return;
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int j = 0; j < NR; ++j) {
      int64_t col = n0 + wc * 16 * NR + j * 16 + (lane & 15);
      float bv = bias ? bias[col] : 0.f;
#pragma unroll
      for (int rgi = 0; rgi < 4; ++rgi) {
        int64_t row = m0 + wr * 128 + i * 16 + (lane >> 4) * 4 + rgi;
        float v = acc[i][j][rgi] + bv;
        if (side) v += (float)((const __bf16*)side)[row * N + col];
        if (FUSE_RELU) v = v > 0.f ? v : 0.f;
        ((float*)C)[row * N + col] = v;
      }
    }
  }
}

}  // namespace

extern "C" {

// Eligibility gate for the 8-phase path (NT, interior-only geometry).
int stf_gemm_bf16_8ph_ok(int64_t M, int64_t N, int64_t K) {
  return (M % 256 == 0) && (N % 64 == 0) && (K % 64 == 0) && K > 0;
}

hipError_t stf_gemm_bf16_8ph_side(const void* A, const void* B, void* C,
                                  const void* bias_f32, const void* side_bf16,
                                  int64_t M, int64_t N, int64_t K, int64_t lda,
                                  int64_t ldb, int out_bf16, int fuse_relu,
                                  hipStream_t stream) {
  if (!stf_gemm_bf16_8ph_ok(M, N, K)) return hipErrorInvalidValue;
  LinearAG ag{(const uint16_t*)A, lda};
  const uint16_t* b = (const uint16_t*)B;
  const float* bias = (const float*)bias_f32;
  const uint16_t* side = (const uint16_t*)side_bf16;
  int nr = (N % 256 == 0) ? 4 : (N % 128 == 0) ? 2 : 1;
  int64_t blocks = (M / kBM) * (N / (64 * nr));
  // LDS: A ring 64 KiB + B ring 2 * (BN*64*2) bytes.
  size_t shmem = 64 * 1024 + 2 * (size_t)(64 * nr) * kBK * 2;
#define STF_8PH_LAUNCH(NR_, OB, FR)                                          \
  hipLaunchKernelGGL((Gemm256x8Ph<NR_, OB, FR, LinearAG>),                   \
                     dim3((uint32_t)blocks), dim3(kThreads), shmem, stream,  \
                     ag, b, C, bias, side, M, N, K, ldb)
#define STF_8PH_NR(NR_)                                                      \
  do {                                                                       \
    if (out_bf16) {                                                          \
      if (fuse_relu) STF_8PH_LAUNCH(NR_, true, true);                        \
      else STF_8PH_LAUNCH(NR_, true, false);                                 \
    } else {                                                                 \
      if (fuse_relu) STF_8PH_LAUNCH(NR_, false, true);                       \
      else STF_8PH_LAUNCH(NR_, false, false);                                \
    }                                                                        \
  } while (0)
  if (nr == 4) STF_8PH_NR(4);
  else if (nr == 2) STF_8PH_NR(2);
  else STF_8PH_NR(1);
#undef STF_8PH_NR
#undef STF_8PH_LAUNCH
  return hipGetLastError();
}

hipError_t stf_gemm_bf16_8ph(const void* A, const void* B, void* C,
                             const void* bias_f32, int64_t M, int64_t N,
                             int64_t K, int64_t lda, int64_t ldb,
                             int out_bf16, int fuse_relu,
                             hipStream_t stream) {
  return stf_gemm_bf16_8ph_side(A, B, C, bias_f32, nullptr, M, N, K, lda,
                                ldb, out_bf16, fuse_relu, stream);
}

// Implicit-GEMM Conv2D forward: y[M=NPQ, N=Cout] = im2col(x) * wt[Cout,rscp]^T
// with the column matrix generated inside the A staging (ConvAG) — the
// MI355X answer to the reference's cuDNN implicit-GEMM conv path
// (conv_ops.cc:664 ThenConvolveWithAlgorithm). Requires C%8==0 (16B loads
// stay within one (r,s) patch) and the 8-phase geometry gate.
hipError_t stf_conv2d_fwd_8ph(const void* x, const void* wt, void* y,
                              const void* bias_f32, const void* zero16,
                              int n, int h, int w, int c, int r, int s_,
                              int sh, int sw, int ph, int pw, int p, int q,
                              int64_t cout, int64_t rscp, int out_bf16,
                              int fuse_relu, hipStream_t stream) {
  int64_t M = (int64_t)n * p * q;
  if (!stf_gemm_bf16_8ph_ok(M, cout, rscp) || (c % 8) != 0)
    return hipErrorInvalidValue;
  ConvAG ag;
  ag.x = (const uint16_t*)x;
  ag.zero16 = (const uint16_t*)zero16;
  ag.div_pq.init((uint32_t)(p * q));
  ag.div_q.init((uint32_t)q);
  ag.div_c.init((uint32_t)c);
  ag.div_s.init((uint32_t)s_);
  ag.H = h; ag.W = w; ag.C = c; ag.S = s_;
  ag.sh = sh; ag.sw = sw; ag.ph = ph; ag.pw = pw;
  ag.rsc = (int64_t)r * s_ * c;
  const uint16_t* b = (const uint16_t*)wt;
  const float* bias = (const float*)bias_f32;
  int nr = (cout % 256 == 0) ? 4 : (cout % 128 == 0) ? 2 : 1;
  int64_t blocks = (M / kBM) * (cout / (64 * nr));
  size_t shmem = 64 * 1024 + 2 * (size_t)(64 * nr) * kBK * 2;
#define STF_C8_LAUNCH(NR_, OB, FR)                                           \
  hipLaunchKernelGGL((Gemm256x8Ph<NR_, OB, FR, ConvAG>),                     \
                     dim3((uint32_t)blocks), dim3(kThreads), shmem, stream,  \
                     ag, b, y, bias, nullptr, M, cout, rscp, rscp)
#define STF_C8_NR(NR_)                                                       \
  do {                                                                       \
    if (out_bf16) {                                                          \
      if (fuse_relu) STF_C8_LAUNCH(NR_, true, true);                         \
      else STF_C8_LAUNCH(NR_, true, false);                                  \
    } else {                                                                 \
      if (fuse_relu) STF_C8_LAUNCH(NR_, false, true);                        \
      else STF_C8_LAUNCH(NR_, false, false);                                 \
    }                                                                        \
  } while (0)
  if (nr == 4) STF_C8_NR(4);
  else if (nr == 2) STF_C8_NR(2);
  else STF_C8_NR(1);
#undef STF_C8_NR
#undef STF_C8_LAUNCH
  return hipGetLastError();
}

}  // extern "C"
