// Shared opcode table + packed-instruction format for the scoped
// elementwise fuser (_FusedElementwise): graph/optimizer.cc emits programs,
// kernels/hip/elementwise.hip (GPU) and kernels/cpu_math.cc (CPU) interpret
// them. Capability analog of the reference's XLA elementwise fusion slot
//  — MI355X-native design: one kernel per fused DAG, the
// intermediate values live in registers, never in HBM.
#pragma once

#include <cstdint>

namespace stf {
namespace fused_ew {

enum Op : int {
  // unary: vals[dst] = f(vals[a])
  kRelu = 1, kRelu6, kSigmoid, kTanh, kExp, kLog, kLog1p, kNeg, kSqrt,
  kRsqrt, kSquare, kAbs, kSoftplus, kSign, kFloor, kReciprocal,
  // binary: vals[dst] = f(vals[a], vals[b])
  kAdd = 64, kSub, kMul, kDiv, kMaximum, kMinimum, kSquaredDifference, kPow,
};

inline bool IsBinary(int op) { return op >= kAdd; }

// instruction: opcode | a<<8 | b<<16 (b unused for unary). Slot space:
// [0, n_inputs) = the op's inputs (slot i broadcasts when input i is a
// scalar); n_inputs + k = result of instruction k.
inline int64_t Pack(int op, int a, int b) {
  return (int64_t)op | ((int64_t)a << 8) | ((int64_t)b << 16);
}
inline void Unpack(int64_t ins, int* op, int* a, int* b) {
  *op = (int)(ins & 0xff);
  *a = (int)((ins >> 8) & 0xff);
  *b = (int)((ins >> 16) & 0xff);
}

constexpr int kMaxInputs = 7;
constexpr int kMaxInstr = 24;

}  // namespace fused_ew
}  // namespace stf
