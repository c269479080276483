// NT bf16 MFMA GEMM for gfx950: C[M,N](f32 or bf16) = A[M,K] * B[N,K]^T.
//
// This is the framework's core matmul/conv engine (replaces the reference's
// cuBLAS path, matmul_op.cc:192 ThenBlasGemm). Structure follows the CDNA4
// guide's "step-3" recipe (§5 ladder): 4-wave workgroup, 128x128 (or
// 128x64/64x64) C tile, K-step 64, double-buffered LDS staged with
// __builtin_amdgcn_global_load_lds (16B), XOR-swizzled LDS image
// (byte ^= (row&7)<<4, both-sides rule 21) read back as ds_read_b128
// fragments feeding v_mfma_f32_16x16x32_bf16, f32 accumulation in AGPRs.
// Interior blocks take the glds fast path; edge blocks (M/N remainder or K
// tail) stage through a guarded scalar path into the same swizzled image.
#include "hip_common.h"

namespace {

__device__ __forceinline__ f32x4 mfma_bf16(bf16x8 a, bf16x8 b, f32x4 c) {
  return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
}

// One operand tile: BR rows x 64 k, stored as 16B chunks with chunk index
// XOR-swizzled by Swz(row). Swz mixes bit 3+ of the row so that row-groups
// of 8 (the K-major staging granularity, whose LDS row stride of 128B
// aliases all 32 banks) still spread their writes across banks.
__device__ __forceinline__ int Swz(int r) { return (r ^ (r >> 3)) & 7; }

template <int BR>
struct TileGeom {
  static constexpr int kSlots = BR * 8;        // 16B slots
  static constexpr int kBytes = kSlots * 16;   // = BR * 128
  static constexpr int kPasses = kSlots / 256; // glds passes (256 threads)
};

// Fast staging: lane-linear glds; the source address carries the inverse
// swizzle so the LDS image is the swizzled one.
template <int BR, class AG>
__device__ __forceinline__ void StageFast(const AG& ag, int row0, int64_t k0,
                                          uint16_t* lds_base, int tid) {
#pragma unroll
  for (int p = 0; p < TileGeom<BR>::kPasses; ++p) {
    int s = p * 256 + tid;
    int r = s >> 3;
    int c8 = (s & 7) ^ Swz(r);
    const uint16_t* g = ag.at(row0 + r, k0 + c8 * 8);
    __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) uint32_t*)g,
                                     (__attribute__((address_space(3))) uint32_t*)(lds_base + (int64_t)s * 8),
                                     16, 0, 0);
  }
}

// K-major staging: the operand is stored [K, R] row-major (contraction dim
// outermost — im2col columns feeding dW, or a weight matrix [K, N] read
// without a separate transpose pass). Each thread loads an 8x8 (k x r)
// block with 16B row loads, transposes it in registers, and writes 8 full
// 16B LDS slots — the Swz(r) swizzle keeps the 8-row-strided writes on
// distinct banks.
template <int BR, class AG>
__device__ __forceinline__ void StageKMajor(const AG& ag, int col0, int64_t k0,
                                            uint16_t* lds_base, int tid) {
  constexpr int kRG = BR / 8;  // 8-wide row groups per k row
  if (tid < 0 || tid >= BR) return;  // one thread per 8x8 block
  int r0 = (tid % kRG) * 8;
  int chunk = tid / kRG;       // k chunk of 8
  uint16_t v[8][8];
#pragma unroll
  for (int kk = 0; kk < 8; ++kk)
    *(ulong2*)v[kk] = *(const ulong2*)ag.at(k0 + chunk * 8 + kk, col0 + r0);
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    uint16_t out[8];
#pragma unroll
    for (int kk = 0; kk < 8; ++kk) out[kk] = v[kk][e];
    int r = r0 + e;
    *(ulong2*)(lds_base + (int64_t)(r * 8 + (chunk ^ Swz(r))) * 8) =
        *(ulong2*)out;
  }
}

template <int BR, class AG>
__device__ __forceinline__ void StageKMajorSafe(
    const AG& ag, int col0, int64_t k0,
    int64_t cols, int64_t K, uint16_t* lds_base, int tid) {
  constexpr int kRG = BR / 8;
  if (tid < 0 || tid >= BR) return;
  int r0 = (tid % kRG) * 8;
  int chunk = tid / kRG;
  uint16_t v[8][8];
#pragma unroll
  for (int kk = 0; kk < 8; ++kk) {
    int64_t krow = k0 + chunk * 8 + kk;
    if (krow < K && col0 + r0 + 8 <= cols) {
      *(ulong2*)v[kk] = *(const ulong2*)ag.at(krow, col0 + r0);
    } else if (krow < K) {
#pragma unroll
      for (int e = 0; e < 8; ++e)
        v[kk][e] = (col0 + r0 + e < cols) ? *ag.at(krow, col0 + r0 + e)
                                          : 0;
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) v[kk][e] = 0;
    }
  }
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    uint16_t out[8];
#pragma unroll
    for (int kk = 0; kk < 8; ++kk) out[kk] = v[kk][e];
    int r = r0 + e;
    *(ulong2*)(lds_base + (int64_t)(r * 8 + (chunk ^ Swz(r))) * 8) =
        *(ulong2*)out;
  }
}

// Guarded staging for edge blocks / K tails: scalar loads, zero padding.
template <int BR, class AG>
__device__ __forceinline__ void StageSafe(const AG& ag, int row0, int64_t k0,
                                          int64_t rows, int64_t K,
                                          uint16_t* lds_base, int tid) {
#pragma unroll
  for (int p = 0; p < TileGeom<BR>::kPasses; ++p) {
    int s = p * 256 + tid;
    int r = s >> 3;
    int c8 = (s & 7) ^ Swz(r);
    uint16_t vals[8];
    int64_t row = row0 + r;
    int64_t kbase = k0 + c8 * 8;
    if (row < rows && kbase + 8 <= K) {
      *(ulong2*)vals = *(const ulong2*)ag.at(row, kbase);
    } else if (row < rows && kbase < K) {
#pragma unroll
      for (int e = 0; e < 8; ++e)
        vals[e] = (kbase + e < K) ? *ag.at(row, kbase + e) : 0;
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) vals[e] = 0;
    }
    *(ulong2*)(lds_base + (int64_t)s * 8) = *(ulong2*)vals;
  }
}

// GEMM kernel template.
//  WAVES_M x WAVES_N waves; each wave computes (WM*16) x (WN*16) outputs.
//  BM = WAVES_M*WM*16, BN = WAVES_N*WN*16, BK = 64.
//  A_KM/B_KM: operand stored contraction-major ([K, M] / [K, N] row-major;
//  lda/ldb = that row stride) and transposed during LDS staging — this is
//  how dW (= col^T · dy) and NN matmuls run without separate transpose
//  kernels. With *_KM false the operand is [M, K] / [N, K] row-major (NT).
// DBUF=false: single-buffered LDS (half the footprint -> 2x the resident
// blocks per CU). Chosen for short K loops (kt <= 4), where double
// buffering cannot warm up and cross-block overlap hides latency better.
template <int WAVES_M, int WAVES_N, int WM, int WN, bool A_KM, bool B_KM,
          bool OUT_BF16, bool FUSE_RELU, bool SPLITK = false,
          bool DBUF = true, class AAG = LinearAG>
__launch_bounds__(256) __global__ void GemmBf16NT(
    AAG a_ag,
    const uint16_t* __restrict__ B,
    void* __restrict__ C,            // [M, N] f32 or bf16 (f32 for SPLITK)
    const float* __restrict__ bias,  // optional [N] f32 bias (nullptr = none)
    int64_t M, int64_t N, int64_t K, int64_t ldb, float beta,
    int splitk = 1) {
  constexpr int BM = WAVES_M * WM * 16;
  constexpr int BN = WAVES_N * WN * 16;
  constexpr int BK = 64;

  __shared__ __attribute__((aligned(16)))
      uint16_t lds[(DBUF ? 2 : 1) * (BM + BN) * BK];
  // buffer layout: [A0][A1][B0][B1]; pick by integer offset (an initialized
  // array of LDS pointers does not compile on gfx950)
  constexpr int NBUF = DBUF ? 2 : 1;
  auto a_tile = [&](int buf) { return lds + (DBUF ? buf : 0) * BM * BK; };
  auto b_tile = [&](int buf) {
    return lds + NBUF * BM * BK + (DBUF ? buf : 0) * BN * BK;
  };

  int nbm = (int)((M + BM - 1) / BM);
  int nbn = (int)((N + BN - 1) / BN);
  int bid = XcdSwizzle(blockIdx.x, nbm * nbn * (SPLITK ? splitk : 1));
  int slice = SPLITK ? bid % splitk : 0;
  if (SPLITK) bid /= splitk;
  int bm = bid / nbn, bn = bid % nbn;
  int64_t m0 = (int64_t)bm * BM, n0 = (int64_t)bn * BN;

  int tid = threadIdx.x;
  int lane = tid & 63;
  int wid = tid >> 6;
  int wr = wid / WAVES_N, wc = wid % WAVES_N;

  bool a_interior = (m0 + BM) <= M;
  bool b_interior = (n0 + BN) <= N;

  f32x4 acc[WM][WN];
#pragma unroll
  for (int i = 0; i < WM; ++i)
#pragma unroll
    for (int j = 0; j < WN; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  int kt_total = (int)((K + BK - 1) / BK);
  int kt_begin = 0, kt_count = kt_total;
  if (SPLITK) {
    int per = (kt_total + splitk - 1) / splitk;
    kt_begin = slice * per;
    kt_count = min(kt_total, kt_begin + per);
    if (kt_begin >= kt_count) return;
  }
  // Stage one K-chunk of both operands into LDS buffer `buf`.
  LinearAG b_ag{B, ldb};
  auto stage = [&](int buf, int64_t k0) {
    bool kfull = k0 + BK <= K;
    if (A_KM) {
      if (a_interior && kfull)
        StageKMajor<BM>(a_ag, (int)m0, k0, a_tile(buf), tid);
      else
        StageKMajorSafe<BM>(a_ag, (int)m0, k0, M, K, a_tile(buf), tid);
    } else {
      if (a_interior && kfull)
        StageFast<BM>(a_ag, (int)m0, k0, a_tile(buf), tid);
      else
        StageSafe<BM>(a_ag, (int)m0, k0, M, K, a_tile(buf), tid);
    }
    if (B_KM) {
      // When A also staged K-major, A used threads [0,BM) — give B the next
      // BN threads so both operand stages run concurrently across waves.
      int bt = A_KM ? tid - BM : tid;
      if (b_interior && kfull)
        StageKMajor<BN>(b_ag, (int)n0, k0, b_tile(buf), bt);
      else
        StageKMajorSafe<BN>(b_ag, (int)n0, k0, N, K, b_tile(buf), bt);
    } else {
      if (b_interior && kfull)
        StageFast<BN>(b_ag, (int)n0, k0, b_tile(buf), tid);
      else
        StageSafe<BN>(b_ag, (int)n0, k0, N, K, b_tile(buf), tid);
    }
  };

  // ---- prologue: stage first tile ----
  stage(0, (int64_t)kt_begin * BK);
  __syncthreads();

  int cur = 0;
  for (int kt = kt_begin; kt < kt_count; ++kt) {
    // issue next tile's loads first (overlap with this tile's compute);
    // single-buffer mode must finish compute before restaging instead.
    if (DBUF && kt + 1 < kt_count) {
      stage(cur ^ 1, (int64_t)(kt + 1) * BK);
    }

    // compute on current tile: 2 mfma K-steps of 32
    const uint16_t* at = a_tile(cur);
    const uint16_t* bt = b_tile(cur);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 afrag[WM], bfrag[WN];
#pragma unroll
      for (int i = 0; i < WM; ++i) {
        int r = wr * WM * 16 + i * 16 + (lane & 15);
        int c8 = (kk * 4 + (lane >> 4)) ^ Swz(r);
        afrag[i] = *(const bf16x8*)(at + (r * 8 + c8) * 8);
      }
#pragma unroll
      for (int j = 0; j < WN; ++j) {
        int r = wc * WN * 16 + j * 16 + (lane & 15);
        int c8 = (kk * 4 + (lane >> 4)) ^ Swz(r);
        bfrag[j] = *(const bf16x8*)(bt + (r * 8 + c8) * 8);
      }
#pragma unroll
      for (int i = 0; i < WM; ++i)
#pragma unroll
        for (int j = 0; j < WN; ++j)
          acc[i][j] = mfma_bf16(afrag[i], bfrag[j], acc[i][j]);
    }
    __syncthreads();
    if (DBUF) {
      cur ^= 1;
    } else if (kt + 1 < kt_count) {
      stage(0, (int64_t)(kt + 1) * BK);
      __syncthreads();
    }
  }

  // ---- epilogue ----
  // Fast path for plain bf16 output on interior tiles: the fragment layout
  // scatters 2-byte stores (quarter-wave 32B segments, ~1.2 TB/s effective
  // on skinny-K streaming GEMMs) — stage the tile through LDS (the A/B
  // buffers are dead now) and emit contiguous 16B stores instead.
  if (OUT_BF16 && !SPLITK && beta == 0.f && a_interior && b_interior &&
      (N & 7) == 0) {
    __syncthreads();  // K-loop reads of LDS are done; reuse as C staging
    uint16_t* cbuf = lds;  // [BM][BN] bf16
#pragma unroll
    for (int i = 0; i < WM; ++i) {
#pragma unroll
      for (int j = 0; j < WN; ++j) {
        int col = wc * WN * 16 + j * 16 + (lane & 15);
        float bv = bias ? bias[n0 + col] : 0.f;
#pragma unroll
        for (int rgi = 0; rgi < 4; ++rgi) {
          int row = wr * WM * 16 + i * 16 + (lane >> 4) * 4 + rgi;
          float v = acc[i][j][rgi] + bv;
          if (FUSE_RELU) v = v > 0.f ? v : 0.f;
          cbuf[row * BN + col] = f32_to_bf16(v);
        }
      }
    }
    __syncthreads();
    constexpr int kChunks = BM * BN / 8;  // 16B chunks
#pragma unroll
    for (int p = 0; p < kChunks / 256; ++p) {
      int sidx = (p * 256 + tid) * 8;
      int r = sidx / BN;
      int c = sidx % BN;
      *(ulong2*)((uint16_t*)C + (m0 + r) * N + n0 + c) =
          *(ulong2*)(cbuf + sidx);
    }
    return;
  }
#pragma unroll
  for (int i = 0; i < WM; ++i) {
#pragma unroll
    for (int j = 0; j < WN; ++j) {
      int64_t col = n0 + wc * WN * 16 + j * 16 + (lane & 15);
      if (col >= N) continue;
      float bv = bias ? bias[col] : 0.f;
#pragma unroll
      for (int rgi = 0; rgi < 4; ++rgi) {
        int64_t row = m0 + wr * WM * 16 + i * 16 + (lane >> 4) * 4 + rgi;
        if (row >= M) continue;
        float v = acc[i][j][rgi] + bv;
        if (FUSE_RELU) v = v > 0.f ? v : 0.f;
        if (SPLITK) {
          atomicAdd((float*)C + row * N + col, v);
          continue;
        }
        if (OUT_BF16) {
          uint16_t* out = (uint16_t*)C + row * N + col;
          if (beta != 0.f) v += beta * bf16_to_f32(*out);
          *out = f32_to_bf16(v);
        } else {
          float* out = (float*)C + row * N + col;
          if (beta != 0.f) v += beta * *out;
          *out = v;
        }
      }
    }
  }
}

template <bool A_KM, bool B_KM, bool OUT_BF16, bool FUSE_RELU>
hipError_t LaunchVariant(const uint16_t* A, const uint16_t* B, void* C,
                         const float* bias, int64_t M, int64_t N, int64_t K,
                         int64_t lda, int64_t ldb, float beta,
                         hipStream_t stream) {
  static const bool no_sb = getenv("STF_GEMM_NO_SB") != nullptr;
  bool short_k = !no_sb && K <= 4 * 64;
  LinearAG a_ag{A, lda};
  auto launch = [&](auto kern, int BM, int BN) {
    int64_t blocks = ((M + BM - 1) / BM) * ((N + BN - 1) / BN);
    hipLaunchKernelGGL(kern, dim3((uint32_t)blocks), dim3(256), 0, stream,
                       a_ag, B, C, bias, M, N, K, ldb, beta, 1);
  };
#define STF_PICK(WM_, WN_, TM, TN)                                           do {                                                                         if (short_k)                                                                 launch(GemmBf16NT<2, 2, WM_, WN_, A_KM, B_KM, OUT_BF16, FUSE_RELU,                           false, false>,                                                  TM, TN);                                                          else                                                                         launch(GemmBf16NT<2, 2, WM_, WN_, A_KM, B_KM, OUT_BF16, FUSE_RELU>,               TM, TN);                                                        } while (0)
  if (M <= 32 && N >= 128) {
    // Skinny-M (RNN batch rows, FC with tiny batch): a 32-row tile spans the
    // whole M in one block row, so B is streamed exactly once; 1x4 wave
    // layout keeps all four waves on distinct N columns.
    if (short_k)
      launch(GemmBf16NT<1, 4, 2, 2, A_KM, B_KM, OUT_BF16, FUSE_RELU, false,
                        false>, 32, 128);
    else
      launch(GemmBf16NT<1, 4, 2, 2, A_KM, B_KM, OUT_BF16, FUSE_RELU>, 32,
             128);
  } else if (N >= 128 && M >= 128) {
    STF_PICK(4, 4, 128, 128);
  } else if (N >= 128) {
    STF_PICK(2, 4, 64, 128);
  } else if (M >= 128) {
    STF_PICK(4, 2, 128, 64);
  } else {
    STF_PICK(2, 2, 64, 64);
  }
#undef STF_PICK
  return hipGetLastError();
}

template <bool A_KM, bool B_KM>
hipError_t LaunchSplitK(const uint16_t* A, const uint16_t* B, float* C,
                        int64_t M, int64_t N, int64_t K, int64_t lda,
                        int64_t ldb, int splitk, hipStream_t stream) {
  LinearAG a_ag{A, lda};
  auto launch = [&](auto kern, int BM, int BN) {
    int64_t blocks = ((M + BM - 1) / BM) * ((N + BN - 1) / BN) * splitk;
    hipLaunchKernelGGL(kern, dim3((uint32_t)blocks), dim3(256), 0, stream,
                       a_ag, B, C, nullptr, M, N, K, ldb, 0.f, splitk);
  };
  if (M <= 32 && N >= 128) {
    launch(GemmBf16NT<1, 4, 2, 2, A_KM, B_KM, false, false, true>, 32, 128);
  } else if (N >= 128 && M >= 128) {
    launch(GemmBf16NT<2, 2, 4, 4, A_KM, B_KM, false, false, true>, 128, 128);
  } else if (N >= 128) {
    launch(GemmBf16NT<2, 2, 2, 4, A_KM, B_KM, false, false, true>, 64, 128);
  } else if (M >= 128) {
    launch(GemmBf16NT<2, 2, 4, 2, A_KM, B_KM, false, false, true>, 128, 64);
  } else {
    launch(GemmBf16NT<2, 2, 2, 2, A_KM, B_KM, false, false, true>, 64, 64);
  }
  return hipGetLastError();
}

}  // namespace

extern "C" int stf_gemm_bf16_8ph_ok(int64_t M, int64_t N, int64_t K);
extern "C" hipError_t stf_gemm_bf16_8ph(const void* A, const void* B, void* C,
                                        const void* bias_f32, int64_t M,
                                        int64_t N, int64_t K, int64_t lda,
                                        int64_t ldb, int out_bf16,
                                        int fuse_relu, hipStream_t stream);

// General entry. a_km/b_km select the contraction-major (transposed-staging)
// path per operand; lda/ldb are row strides of the stored layouts
// (a_km ? [K,M] : [M,K], b_km ? [K,N] : [N,K]).
extern "C" hipError_t stf_gemm_bf16(const void* A, const void* B, void* C,
                                    const void* bias_f32, int64_t M, int64_t N,
                                    int64_t K, int64_t lda, int64_t ldb,
                                    float beta, int a_km, int b_km,
                                    int out_bf16, int fuse_relu,
                                    hipStream_t stream) {
  // Fat NT tiles take the 256² 8-phase template (guide §5 top rung);
  // STF_NO_8PH reverts to the 128² step-3 kernel for A/B comparison.
  static const bool no_8ph = getenv("STF_NO_8PH") != nullptr;
  if (!no_8ph && !a_km && !b_km && beta == 0.f &&
      stf_gemm_bf16_8ph_ok(M, N, K))
    return stf_gemm_bf16_8ph(A, B, C, bias_f32, M, N, K, lda, ldb, out_bf16,
                             fuse_relu, stream);
  const uint16_t* a = (const uint16_t*)A;
  const uint16_t* b = (const uint16_t*)B;
  const float* bias = (const float*)bias_f32;
  int sel = (a_km ? 8 : 0) | (b_km ? 4 : 0) | (out_bf16 ? 2 : 0) |
            (fuse_relu ? 1 : 0);
  switch (sel) {
#define STF_CASE(AK, BK_, OB, FR)                                         \
  case ((AK ? 8 : 0) | (BK_ ? 4 : 0) | (OB ? 2 : 0) | (FR ? 1 : 0)):      \
    return LaunchVariant<AK, BK_, OB, FR>(a, b, C, bias, M, N, K, lda,    \
                                          ldb, beta, stream)
    STF_CASE(false, false, false, false);
    STF_CASE(false, false, false, true);
    STF_CASE(false, false, true, false);
    STF_CASE(false, false, true, true);
    STF_CASE(false, true, false, false);
    STF_CASE(false, true, false, true);
    STF_CASE(false, true, true, false);
    STF_CASE(false, true, true, true);
    STF_CASE(true, false, false, false);
    STF_CASE(true, false, false, true);
    STF_CASE(true, false, true, false);
    STF_CASE(true, false, true, true);
    STF_CASE(true, true, false, false);
    STF_CASE(true, true, false, true);
    STF_CASE(true, true, true, false);
    STF_CASE(true, true, true, true);
#undef STF_CASE
  }
  return hipErrorInvalidValue;
}

// Split-K for skinny-output huge-K GEMMs (dW): partial products atomically
// accumulated into a zeroed f32 C (guide split-K decomposition: size the
// grid to ~1-2x the 256 CUs).
extern "C" hipError_t stf_gemm_bf16_splitk(const void* A, const void* B,
                                           void* C_f32, int64_t M, int64_t N,
                                           int64_t K, int64_t lda, int64_t ldb,
                                           int a_km, int b_km, int splitk,
                                           hipStream_t stream) {
  const uint16_t* a = (const uint16_t*)A;
  const uint16_t* b = (const uint16_t*)B;
  float* c = (float*)C_f32;
  if (a_km && b_km)
    return LaunchSplitK<true, true>(a, b, c, M, N, K, lda, ldb, splitk,
                                    stream);
  if (a_km)
    return LaunchSplitK<true, false>(a, b, c, M, N, K, lda, ldb, splitk,
                                     stream);
  if (b_km)
    return LaunchSplitK<false, true>(a, b, c, M, N, K, lda, ldb, splitk,
                                     stream);
  return LaunchSplitK<false, false>(a, b, c, M, N, K, lda, ldb, splitk,
                                    stream);
}

// Legacy NT entries (both operands row-major with contraction innermost).
extern "C" hipError_t stf_gemm_bf16_nt_splitk(const void* A, const void* B,
                                              void* C_f32, int64_t M,
                                              int64_t N, int64_t K, int splitk,
                                              hipStream_t stream) {
  return stf_gemm_bf16_splitk(A, B, C_f32, M, N, K, K, K, 0, 0, splitk,
                              stream);
}

// C[M,N] = A[M,K] * B[N,K]^T (+ beta*C) (+bias) (+relu)
extern "C" hipError_t stf_gemm_bf16_nt(const void* A, const void* B, void* C,
                                       const void* bias_f32, int64_t M,
                                       int64_t N, int64_t K, float beta,
                                       int out_bf16, int fuse_relu,
                                       hipStream_t stream) {
  return stf_gemm_bf16(A, B, C, bias_f32, M, N, K, K, K, beta, 0, 0,
                       out_bf16, fuse_relu, stream);
}

// Implicit-GEMM Conv2D backprop-filter: dW[rsc, Cout] = im2col(x)^T * dy,
// contraction over the NPQ output pixels, split-K with f32 atomics. The
// column matrix is generated inside the K-major A staging (ConvAG) — the
// materialized im2col buffer and its cache disappear (reference analog:
// conv_grad_filter_ops.cc ThenConvolveBackwardFilterWithAlgorithm).
extern "C" hipError_t stf_conv2d_dw_splitk(
    const void* x, const void* dy, void* dw_f32, const void* zero16, int n,
    int h, int w, int c, int r, int s_, int sh, int sw, int ph, int pw,
    int p, int q, int64_t cout, int splitk, hipStream_t stream) {
  if ((c % 8) != 0) return hipErrorInvalidValue;
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
  int64_t M = ag.rsc;            // dW rows
  int64_t N = cout;
  int64_t K = (int64_t)n * p * q;  // contraction: output pixels
  const uint16_t* b = (const uint16_t*)dy;
  float* cptr = (float*)dw_f32;
  auto launch = [&](auto kern, int BM, int BN) {
    int64_t blocks = ((M + BM - 1) / BM) * ((N + BN - 1) / BN) * splitk;
    hipLaunchKernelGGL(kern, dim3((uint32_t)blocks), dim3(256), 0, stream,
                       ag, b, cptr, nullptr, M, N, K, cout, 0.f, splitk);
  };
  if (N >= 128 && M >= 128) {
    launch(GemmBf16NT<2, 2, 4, 4, true, true, false, false, true, true,
                      ConvAG>, 128, 128);
  } else if (N >= 128) {
    launch(GemmBf16NT<2, 2, 2, 4, true, true, false, false, true, true,
                      ConvAG>, 64, 128);
  } else if (M >= 128) {
    launch(GemmBf16NT<2, 2, 4, 2, true, true, false, false, true, true,
                      ConvAG>, 128, 64);
  } else {
    launch(GemmBf16NT<2, 2, 2, 2, true, true, false, false, true, true,
                      ConvAG>, 64, 64);
  }
  return hipGetLastError();
}
// Implicit-GEMM Conv2D forward through the general NT template: the column
// matrix row (one output pixel's [r,s,c] window) is generated inside the A
// staging by ConvAG — covers the shapes the 256-row 8-phase kernel cannot
// take (couts or M not multiples of its tile). Weights stay [rsc, K]
// contraction-major (B_KM staging transposes in-register).
extern "C" hipError_t stf_conv2d_fwd_nt(
    const void* x, const void* w, void* y, const void* zero16, int n, int h,
    int w_, int c, int r, int s_, int sh, int sw, int ph, int pw, int p,
    int q, int64_t cout, int64_t rscp, hipStream_t stream) {
  if ((c % 8) != 0) return hipErrorInvalidValue;
  ConvAG ag;
  ag.x = (const uint16_t*)x;
  ag.zero16 = (const uint16_t*)zero16;
  ag.div_pq.init((uint32_t)(p * q));
  ag.div_q.init((uint32_t)q);
  ag.div_c.init((uint32_t)c);
  ag.div_s.init((uint32_t)s_);
  ag.H = h; ag.W = w_; ag.C = c; ag.S = s_;
  ag.sh = sh; ag.sw = sw; ag.ph = ph; ag.pw = pw;
  ag.rsc = (int64_t)r * s_ * c;
  int64_t M = (int64_t)n * p * q;
  int64_t N = cout;
  int64_t K = rscp;
  const uint16_t* b = (const uint16_t*)w;
  auto launch = [&](auto kern, int BM, int BN) {
    int64_t blocks = ((M + BM - 1) / BM) * ((N + BN - 1) / BN);
    hipLaunchKernelGGL(kern, dim3((uint32_t)blocks), dim3(256), 0, stream,
                       ag, b, y, nullptr, M, N, K, cout, 0.f, 1);
  };
  if (N >= 128 && M >= 128) {
    launch(GemmBf16NT<2, 2, 4, 4, false, true, true, false, false, true,
                      ConvAG>, 128, 128);
  } else if (N >= 128) {
    launch(GemmBf16NT<2, 2, 2, 4, false, true, true, false, false, true,
                      ConvAG>, 64, 128);
  } else if (M >= 128) {
    launch(GemmBf16NT<2, 2, 4, 2, false, true, true, false, false, true,
                      ConvAG>, 128, 64);
  } else {
    launch(GemmBf16NT<2, 2, 2, 2, false, true, true, false, false, true,
                      ConvAG>, 64, 64);
  }
  return hipGetLastError();
}

