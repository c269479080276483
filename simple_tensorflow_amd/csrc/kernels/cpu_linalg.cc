// Batched dense linear algebra CPU kernels (capability analogs of the
// reference's core/kernels/cholesky_op.cc, determinant_op.cc,
// matrix_inverse_op.cc, matrix_solve_op.cc, matrix_triangular_solve_op.cc,
// qr_op_impl.h, svd_op_impl.h, self_adjoint_eig_v2_op_impl.h — the reference
// delegates to Eigen; these are self-contained implementations: LU with
// partial pivoting, Cholesky, Householder QR, one-sided Jacobi SVD and
// cyclic Jacobi symmetric eigendecomposition, all accumulating in double).
#include <algorithm>
#include <cmath>
#include <vector>

#include "framework/op_kernel.h"

namespace stf {
namespace {

using Mat = std::vector<double>;  // row-major m x n

// ---- helpers -------------------------------------------------------------

// LU factorization with partial pivoting in place. Returns sign of the
// permutation, 0 if singular. piv[i] = pivot row chosen at step i.
int LuFactor(Mat& a, int n, std::vector<int>& piv) {
  int sign = 1;
  piv.resize(n);
  for (int k = 0; k < n; ++k) {
    int p = k;
    double mx = std::fabs(a[k * n + k]);
    for (int i = k + 1; i < n; ++i) {
      double v = std::fabs(a[i * n + k]);
      if (v > mx) {
        mx = v;
        p = i;
      }
    }
    piv[k] = p;
    if (mx == 0.0) return 0;
    if (p != k) {
      for (int j = 0; j < n; ++j) std::swap(a[k * n + j], a[p * n + j]);
      sign = -sign;
    }
    double d = a[k * n + k];
    for (int i = k + 1; i < n; ++i) {
      double f = a[i * n + k] / d;
      a[i * n + k] = f;
      for (int j = k + 1; j < n; ++j) a[i * n + j] -= f * a[k * n + j];
    }
  }
  return sign;
}

// Solve LU x = b for one rhs column vector (b modified in place). The row
// interchanges must ALL be applied before forward substitution (LAPACK
// dlaswp order): the stored multipliers sit at their final row positions,
// so replay the permutation history first.
void LuSolveVec(const Mat& lu, int n, const std::vector<int>& piv,
                double* b) {
  for (int k = 0; k < n; ++k)
    if (piv[k] != k) std::swap(b[k], b[piv[k]]);
  for (int k = 0; k < n; ++k) {
    for (int i = k + 1; i < n; ++i) b[i] -= lu[i * n + k] * b[k];
  }
  for (int i = n - 1; i >= 0; --i) {
    for (int j = i + 1; j < n; ++j) b[i] -= lu[i * n + j] * b[j];
    b[i] /= lu[i * n + i];
  }
}

// Cholesky in place -> lower triangular (upper zeroed). false if not SPD.
bool CholeskyFactor(Mat& a, int n) {
  for (int j = 0; j < n; ++j) {
    double d = a[j * n + j];
    for (int k = 0; k < j; ++k) d -= a[j * n + k] * a[j * n + k];
    if (d <= 0.0) return false;
    d = std::sqrt(d);
    a[j * n + j] = d;
    for (int i = j + 1; i < n; ++i) {
      double v = a[i * n + j];
      for (int k = 0; k < j; ++k) v -= a[i * n + k] * a[j * n + k];
      a[i * n + j] = v / d;
    }
    for (int k = j + 1; k < n; ++k) a[j * n + k] = 0.0;
  }
  return true;
}

// Householder QR: a (m x n, m >= n) -> q (m x kq), r (kq_r x n) with
// kq = full ? m : n. Straightforward accumulation of reflectors.
void QrFactor(const Mat& a_in, int m, int n, bool full, Mat& q, Mat& r) {
  Mat a = a_in;  // working copy, becomes R
  int kmin = std::min(m, n);
  int kq = full ? m : kmin;
  // accumulate Q as product of reflectors applied to identity
  q.assign((size_t)m * kq, 0.0);
  Mat qfull((size_t)m * m, 0.0);
  for (int i = 0; i < m; ++i) qfull[i * m + i] = 1.0;
  std::vector<double> v(m);
  for (int k = 0; k < kmin; ++k) {
    double norm = 0.0;
    for (int i = k; i < m; ++i) norm += a[i * n + k] * a[i * n + k];
    norm = std::sqrt(norm);
    if (norm == 0.0) continue;
    double alpha = a[k * n + k] >= 0 ? -norm : norm;
    double vnorm2 = 0.0;
    for (int i = k; i < m; ++i) {
      v[i] = a[i * n + k] - (i == k ? alpha : 0.0);
      vnorm2 += v[i] * v[i];
    }
    if (vnorm2 == 0.0) continue;
    // apply H = I - 2 v v^T / vnorm2 to A (cols k..n)
    for (int j = k; j < n; ++j) {
      double dot = 0.0;
      for (int i = k; i < m; ++i) dot += v[i] * a[i * n + j];
      double f = 2.0 * dot / vnorm2;
      for (int i = k; i < m; ++i) a[i * n + j] -= f * v[i];
    }
    // apply to Qfull from the right: Qfull = Qfull * H
    for (int i = 0; i < m; ++i) {
      double dot = 0.0;
      for (int j = k; j < m; ++j) dot += qfull[i * m + j] * v[j];
      double f = 2.0 * dot / vnorm2;
      for (int j = k; j < m; ++j) qfull[i * m + j] -= f * v[j];
    }
  }
  // sign convention: make R's diagonal non-negative
  for (int k = 0; k < kmin; ++k) {
    if (a[k * n + k] < 0.0) {
      for (int j = k; j < n; ++j) a[k * n + j] = -a[k * n + j];
      for (int i = 0; i < m; ++i) qfull[i * m + k] = -qfull[i * m + k];
    }
  }
  for (int i = 0; i < m; ++i)
    for (int j = 0; j < kq; ++j) q[i * kq + j] = qfull[i * m + j];
  int kr = full ? m : kmin;
  r.assign((size_t)kr * n, 0.0);
  for (int i = 0; i < std::min(kr, m); ++i)
    for (int j = i; j < n; ++j) r[i * n + j] = a[i * n + j];
}

// Cyclic Jacobi eigendecomposition of symmetric a (n x n): a = V diag(e) V^T.
// e ascending; V columns are eigenvectors.
void JacobiEig(Mat a, int n, std::vector<double>& e, Mat& v) {
  v.assign((size_t)n * n, 0.0);
  for (int i = 0; i < n; ++i) v[i * n + i] = 1.0;
  for (int sweep = 0; sweep < 60; ++sweep) {
    double off = 0.0;
    for (int p = 0; p < n; ++p)
      for (int q = p + 1; q < n; ++q) off += a[p * n + q] * a[p * n + q];
    if (off < 1e-30) break;
    for (int p = 0; p < n; ++p) {
      for (int q = p + 1; q < n; ++q) {
        double apq = a[p * n + q];
        if (std::fabs(apq) < 1e-300) continue;
        double app = a[p * n + p], aqq = a[q * n + q];
        double tau = (aqq - app) / (2.0 * apq);
        double t = (tau >= 0 ? 1.0 : -1.0) /
                   (std::fabs(tau) + std::sqrt(1.0 + tau * tau));
        double c = 1.0 / std::sqrt(1.0 + t * t), s = t * c;
        for (int k = 0; k < n; ++k) {
          double akp = a[k * n + p], akq = a[k * n + q];
          a[k * n + p] = c * akp - s * akq;
          a[k * n + q] = s * akp + c * akq;
        }
        for (int k = 0; k < n; ++k) {
          double apk = a[p * n + k], aqk = a[q * n + k];
          a[p * n + k] = c * apk - s * aqk;
          a[q * n + k] = s * apk + c * aqk;
        }
        for (int k = 0; k < n; ++k) {
          double vkp = v[k * n + p], vkq = v[k * n + q];
          v[k * n + p] = c * vkp - s * vkq;
          v[k * n + q] = s * vkp + c * vkq;
        }
      }
    }
  }
  e.resize(n);
  for (int i = 0; i < n; ++i) e[i] = a[i * n + i];
  // sort ascending with eigenvector columns
  std::vector<int> idx(n);
  for (int i = 0; i < n; ++i) idx[i] = i;
  std::sort(idx.begin(), idx.end(),
            [&](int x, int y) { return e[x] < e[y]; });
  std::vector<double> es(n);
  Mat vs((size_t)n * n);
  for (int j = 0; j < n; ++j) {
    es[j] = e[idx[j]];
    for (int i = 0; i < n; ++i) vs[i * n + j] = v[i * n + idx[j]];
  }
  e = es;
  v = vs;
}

struct BatchView {
  int64_t nbatch, m, n;
};

Status GetBatch(const Tensor& t, BatchView* bv, int min_rank = 2) {
  int dims = t.shape().dims();
  if (dims < min_rank)
    return errors::InvalidArgument("expected rank >= 2, got ", dims);
  bv->m = t.shape().dim_size(dims - 2);
  bv->n = t.shape().dim_size(dims - 1);
  bv->nbatch = 1;
  for (int i = 0; i < dims - 2; ++i) bv->nbatch *= t.shape().dim_size(i);
  return Status::OK();
}

template <typename T>
void LoadMat(const T* src, Mat& dst, int64_t count) {
  dst.resize(count);
  for (int64_t i = 0; i < count; ++i) dst[i] = (double)src[i];
}

template <typename T>
void StoreMat(const Mat& src, T* dst, int64_t count) {
  for (int64_t i = 0; i < count; ++i) dst[i] = (T)src[i];
}

// ---- kernels -------------------------------------------------------------

template <typename T>
class CholeskyOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    BatchView bv;
    Status s = GetBatch(in, &bv);
    if (!s.ok() || bv.m != bv.n) {
      ctx->SetStatus(errors::InvalidArgument("Cholesky: square input required"));
      return;
    }
    Tensor* out = ctx->allocate_output(0, in.shape());
    int n = (int)bv.n;
    Mat a;
    for (int64_t b = 0; b < bv.nbatch; ++b) {
      LoadMat(in.flat<T>() + b * n * n, a, (int64_t)n * n);
      if (!CholeskyFactor(a, n)) {
        ctx->SetStatus(errors::InvalidArgument(
            "Cholesky: input is not positive definite"));
        return;
      }
      StoreMat(a, out->flat<T>() + b * n * n, (int64_t)n * n);
    }
  }
};

// Reverse-mode Cholesky gradient (Iain Murray 2016 "Differentiation of the
// Cholesky decomposition", blocked level-2 variant of the reference's
// cholesky_grad.cc).
template <typename T>
class CholeskyGradOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& lt = ctx->input(0);
    const Tensor& gt = ctx->input(1);
    BatchView bv;
    Status s = GetBatch(lt, &bv);
    if (!s.ok() || bv.m != bv.n) {
      ctx->SetStatus(errors::InvalidArgument("CholeskyGrad: bad input"));
      return;
    }
    int n = (int)bv.n;
    Tensor* out = ctx->allocate_output(0, lt.shape());
    Mat L, Lbar;
    for (int64_t b = 0; b < bv.nbatch; ++b) {
      LoadMat(lt.flat<T>() + b * n * n, L, (int64_t)n * n);
      LoadMat(gt.flat<T>() + b * n * n, Lbar, (int64_t)n * n);
      // zero the upper triangle of the incoming grad
      for (int i = 0; i < n; ++i)
        for (int j = i + 1; j < n; ++j) Lbar[i * n + j] = 0.0;
      // unblocked reverse-mode recurrence
      for (int k = n - 1; k >= 0; --k) {
        for (int j = k + 1; j < n; ++j)
          for (int i = j; i < n; ++i) {
            Lbar[i * n + k] -= Lbar[i * n + j] * L[j * n + k];
            Lbar[j * n + k] -= Lbar[i * n + j] * L[i * n + k];
          }
        for (int j = k + 1; j < n; ++j)
          Lbar[j * n + k] /= L[k * n + k];
        double d = Lbar[k * n + k];
        for (int j = k + 1; j < n; ++j)
          d -= Lbar[j * n + k] * L[j * n + k];
        Lbar[k * n + k] = 0.5 * d / L[k * n + k];
      }
      // symmetrize: dA = (Lbar + Lbar^T)/2 with full weight on diagonal
      Mat dA((size_t)n * n, 0.0);
      for (int i = 0; i < n; ++i)
        for (int j = 0; j <= i; ++j) {
          double v = Lbar[i * n + j] * (i == j ? 1.0 : 0.5);
          dA[i * n + j] = v;
          dA[j * n + i] = v;
        }
      StoreMat(dA, out->flat<T>() + b * n * n, (int64_t)n * n);
    }
  }
};

template <typename T>
class MatrixDeterminantOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    BatchView bv;
    Status s = GetBatch(in, &bv);
    if (!s.ok() || bv.m != bv.n) {
      ctx->SetStatus(errors::InvalidArgument("det: square input required"));
      return;
    }
    TensorShape out_shape;
    for (int i = 0; i < in.shape().dims() - 2; ++i)
      out_shape.AddDim(in.shape().dim_size(i));
    Tensor* out = ctx->allocate_output(0, out_shape);
    int n = (int)bv.n;
    Mat a;
    std::vector<int> piv;
    for (int64_t b = 0; b < bv.nbatch; ++b) {
      LoadMat(in.flat<T>() + b * n * n, a, (int64_t)n * n);
      int sign = LuFactor(a, n, piv);
      double det = sign;
      for (int i = 0; i < n; ++i) det *= a[i * n + i];
      if (sign == 0) det = 0.0;
      out->flat<T>()[b] = (T)det;
    }
  }
};

template <typename T>
class MatrixInverseOp : public OpKernel {
 public:
  explicit MatrixInverseOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("adjoint", &adjoint_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    BatchView bv;
    Status s = GetBatch(in, &bv);
    if (!s.ok() || bv.m != bv.n) {
      ctx->SetStatus(errors::InvalidArgument("inverse: square input required"));
      return;
    }
    Tensor* out = ctx->allocate_output(0, in.shape());
    int n = (int)bv.n;
    Mat a;
    std::vector<int> piv;
    std::vector<double> col(n);
    for (int64_t b = 0; b < bv.nbatch; ++b) {
      LoadMat(in.flat<T>() + b * n * n, a, (int64_t)n * n);
      if (adjoint_) {
        for (int i = 0; i < n; ++i)
          for (int j = i + 1; j < n; ++j) std::swap(a[i * n + j], a[j * n + i]);
      }
      if (LuFactor(a, n, piv) == 0) {
        ctx->SetStatus(errors::InvalidArgument("inverse: singular matrix"));
        return;
      }
      T* op = out->flat<T>() + b * n * n;
      for (int j = 0; j < n; ++j) {
        std::fill(col.begin(), col.end(), 0.0);
        col[j] = 1.0;
        LuSolveVec(a, n, piv, col.data());
        for (int i = 0; i < n; ++i) op[i * n + j] = (T)col[i];
      }
    }
  }

 private:
  bool adjoint_ = false;
};

template <typename T>
class MatrixSolveOp : public OpKernel {
 public:
  explicit MatrixSolveOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("adjoint", &adjoint_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& mt = ctx->input(0);
    const Tensor& rt = ctx->input(1);
    BatchView bm, br;
    Status s1 = GetBatch(mt, &bm), s2 = GetBatch(rt, &br);
    if (!s1.ok() || !s2.ok() || bm.m != bm.n || br.m != bm.n ||
        bm.nbatch != br.nbatch) {
      ctx->SetStatus(errors::InvalidArgument("solve: incompatible shapes"));
      return;
    }
    Tensor* out = ctx->allocate_output(0, rt.shape());
    int n = (int)bm.n, k = (int)br.n;
    Mat a;
    std::vector<int> piv;
    std::vector<double> col(n);
    for (int64_t b = 0; b < bm.nbatch; ++b) {
      LoadMat(mt.flat<T>() + b * n * n, a, (int64_t)n * n);
      if (adjoint_) {
        for (int i = 0; i < n; ++i)
          for (int j = i + 1; j < n; ++j) std::swap(a[i * n + j], a[j * n + i]);
      }
      if (LuFactor(a, n, piv) == 0) {
        ctx->SetStatus(errors::InvalidArgument("solve: singular matrix"));
        return;
      }
      const T* rp = rt.flat<T>() + b * n * k;
      T* op = out->flat<T>() + b * n * k;
      for (int j = 0; j < k; ++j) {
        for (int i = 0; i < n; ++i) col[i] = (double)rp[i * k + j];
        LuSolveVec(a, n, piv, col.data());
        for (int i = 0; i < n; ++i) op[i * k + j] = (T)col[i];
      }
    }
  }

 private:
  bool adjoint_ = false;
};

template <typename T>
class MatrixTriangularSolveOp : public OpKernel {
 public:
  explicit MatrixTriangularSolveOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("lower", &lower_);
    c->GetAttr("adjoint", &adjoint_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& mt = ctx->input(0);
    const Tensor& rt = ctx->input(1);
    BatchView bm, br;
    Status s1 = GetBatch(mt, &bm), s2 = GetBatch(rt, &br);
    if (!s1.ok() || !s2.ok() || bm.m != bm.n || br.m != bm.n ||
        bm.nbatch != br.nbatch) {
      ctx->SetStatus(errors::InvalidArgument("tri-solve: bad shapes"));
      return;
    }
    Tensor* out = ctx->allocate_output(0, rt.shape());
    int n = (int)bm.n, k = (int)br.n;
    // effective orientation: adjoint of lower behaves like upper
    bool low = adjoint_ ? !lower_ : lower_;
    Mat a;
    for (int64_t b = 0; b < bm.nbatch; ++b) {
      LoadMat(mt.flat<T>() + b * n * n, a, (int64_t)n * n);
      if (adjoint_) {
        for (int i = 0; i < n; ++i)
          for (int j = i + 1; j < n; ++j) std::swap(a[i * n + j], a[j * n + i]);
      }
      const T* rp = rt.flat<T>() + b * n * k;
      T* op = out->flat<T>() + b * n * k;
      for (int j = 0; j < k; ++j) {
        if (low) {
          for (int i = 0; i < n; ++i) {
            double v = (double)rp[i * k + j];
            for (int l = 0; l < i; ++l) v -= a[i * n + l] * (double)op[l * k + j];
            op[i * k + j] = (T)(v / a[i * n + i]);
          }
        } else {
          for (int i = n - 1; i >= 0; --i) {
            double v = (double)rp[i * k + j];
            for (int l = i + 1; l < n; ++l)
              v -= a[i * n + l] * (double)op[l * k + j];
            op[i * k + j] = (T)(v / a[i * n + i]);
          }
        }
      }
    }
  }

 private:
  bool lower_ = true, adjoint_ = false;
};

template <typename T>
class QrOp : public OpKernel {
 public:
  explicit QrOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("full_matrices", &full_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    BatchView bv;
    Status s = GetBatch(in, &bv);
    if (!s.ok()) {
      ctx->SetStatus(s);
      return;
    }
    int m = (int)bv.m, n = (int)bv.n;
    int kmin = std::min(m, n);
    int kq = full_ ? m : kmin;
    int kr = full_ ? m : kmin;
    TensorShape qs, rs;
    for (int i = 0; i < in.shape().dims() - 2; ++i) {
      qs.AddDim(in.shape().dim_size(i));
      rs.AddDim(in.shape().dim_size(i));
    }
    qs.AddDim(m);
    qs.AddDim(kq);
    rs.AddDim(kr);
    rs.AddDim(n);
    Tensor* qt = ctx->allocate_output(0, qs);
    Tensor* rt = ctx->allocate_output(1, rs);
    Mat a, q, r;
    for (int64_t b = 0; b < bv.nbatch; ++b) {
      LoadMat(in.flat<T>() + b * m * n, a, (int64_t)m * n);
      QrFactor(a, m, n, full_, q, r);
      StoreMat(q, qt->flat<T>() + b * m * kq, (int64_t)m * kq);
      StoreMat(r, rt->flat<T>() + b * kr * n, (int64_t)kr * n);
    }
  }

 private:
  bool full_ = false;
};

template <typename T>
class SelfAdjointEigV2Op : public OpKernel {
 public:
  explicit SelfAdjointEigV2Op(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("compute_v", &compute_v_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    BatchView bv;
    Status s = GetBatch(in, &bv);
    if (!s.ok() || bv.m != bv.n) {
      ctx->SetStatus(errors::InvalidArgument("eig: square input required"));
      return;
    }
    int n = (int)bv.n;
    TensorShape es;
    for (int i = 0; i < in.shape().dims() - 2; ++i)
      es.AddDim(in.shape().dim_size(i));
    es.AddDim(n);
    Tensor* et = ctx->allocate_output(0, es);
    Tensor* vt = ctx->allocate_output(
        1, compute_v_ ? in.shape()
                      : TensorShape({0}));
    Mat a, v;
    std::vector<double> e;
    for (int64_t b = 0; b < bv.nbatch; ++b) {
      LoadMat(in.flat<T>() + b * n * n, a, (int64_t)n * n);
      JacobiEig(a, n, e, v);
      for (int i = 0; i < n; ++i) et->flat<T>()[b * n + i] = (T)e[i];
      if (compute_v_)
        StoreMat(v, vt->flat<T>() + b * n * n, (int64_t)n * n);
    }
  }

 private:
  bool compute_v_ = true;
};

// SVD via symmetric eig of A^T A (n x n side; fine for the sizes these CPU
// kernels serve). u = A v / s with Gram-Schmidt completion for full_matrices.
template <typename T>
class SvdOp : public OpKernel {
 public:
  explicit SvdOp(OpKernelConstruction* c) : OpKernel(c) {
    c->GetAttr("compute_uv", &compute_uv_);
    c->GetAttr("full_matrices", &full_);
  }
  void Compute(OpKernelContext* ctx) override {
    const Tensor& in = ctx->input(0);
    BatchView bv;
    Status st = GetBatch(in, &bv);
    if (!st.ok()) {
      ctx->SetStatus(st);
      return;
    }
    int m = (int)bv.m, n = (int)bv.n;
    bool transposed = m < n;  // compute on the tall orientation
    int tm = transposed ? n : m, tn = transposed ? m : n;
    int p = tn;  // = min(m, n)
    TensorShape ss, us, vs;
    for (int i = 0; i < in.shape().dims() - 2; ++i) {
      ss.AddDim(in.shape().dim_size(i));
      us.AddDim(in.shape().dim_size(i));
      vs.AddDim(in.shape().dim_size(i));
    }
    ss.AddDim(p);
    int ucols = full_ ? m : p, vcols = full_ ? n : p;
    us.AddDim(m);
    us.AddDim(ucols);
    vs.AddDim(n);
    vs.AddDim(vcols);
    Tensor* stt = ctx->allocate_output(0, ss);
    Tensor* ut = ctx->allocate_output(1, compute_uv_ ? us : TensorShape({0}));
    Tensor* vt = ctx->allocate_output(2, compute_uv_ ? vs : TensorShape({0}));
    Mat a((size_t)tm * tn), ata((size_t)tn * tn), evec;
    std::vector<double> eval;
    for (int64_t b = 0; b < bv.nbatch; ++b) {
      const T* src = in.flat<T>() + b * m * n;
      for (int i = 0; i < tm; ++i)
        for (int j = 0; j < tn; ++j)
          a[i * tn + j] =
              transposed ? (double)src[j * n + i] : (double)src[i * n + j];
      // A^T A
      for (int i = 0; i < tn; ++i)
        for (int j = 0; j < tn; ++j) {
          double acc = 0.0;
          for (int k = 0; k < tm; ++k) acc += a[k * tn + i] * a[k * tn + j];
          ata[i * tn + j] = acc;
        }
      JacobiEig(ata, tn, eval, evec);  // ascending
      // descending singular values
      std::vector<double> sv(p);
      Mat V((size_t)tn * p);  // right singular vectors of the tall A
      for (int j = 0; j < p; ++j) {
        double ev = eval[tn - 1 - j];
        sv[j] = ev > 0 ? std::sqrt(ev) : 0.0;
        for (int i = 0; i < tn; ++i)
          V[i * p + j] = evec[i * tn + (tn - 1 - j)];
      }
      for (int j = 0; j < p; ++j) stt->flat<T>()[b * p + j] = (T)sv[j];
      if (!compute_uv_) continue;
      // U (tall side) = A V / s, Gram-Schmidt for tiny/zero singulars
      Mat U((size_t)tm * p, 0.0);
      for (int j = 0; j < p; ++j) {
        if (sv[j] > 1e-290) {
          for (int i = 0; i < tm; ++i) {
            double acc = 0.0;
            for (int k = 0; k < tn; ++k) acc += a[i * tn + k] * V[k * p + j];
            U[i * p + j] = acc / sv[j];
          }
        }
      }
      auto write_uv = [&](Mat& tallU, Mat& smallV) {
        // map back to original orientation
        T* up = ut->flat<T>() + b * m * ucols;
        T* vp = vt->flat<T>() + b * n * vcols;
        // zero (handles full_matrices padding columns)
        for (int64_t i = 0; i < (int64_t)m * ucols; ++i) up[i] = (T)0;
        for (int64_t i = 0; i < (int64_t)n * vcols; ++i) vp[i] = (T)0;
        if (!transposed) {
          for (int i = 0; i < m; ++i)
            for (int j = 0; j < p; ++j) up[i * ucols + j] = (T)tallU[i * p + j];
          for (int i = 0; i < n; ++i)
            for (int j = 0; j < p; ++j) vp[i * vcols + j] = (T)smallV[i * p + j];
        } else {
          // original A = (tall A)^T: U_orig = V_tall, V_orig = U_tall
          for (int i = 0; i < m; ++i)
            for (int j = 0; j < p; ++j) up[i * ucols + j] = (T)smallV[i * p + j];
          for (int i = 0; i < n; ++i)
            for (int j = 0; j < p; ++j) vp[i * vcols + j] = (T)tallU[i * p + j];
        }
      };
      write_uv(U, V);
    }
  }

 private:
  bool compute_uv_ = true, full_ = false;
};

template <typename T>
class MatrixSolveLsOp : public OpKernel {
 public:
  using OpKernel::OpKernel;
  void Compute(OpKernelContext* ctx) override {
    const Tensor& mt = ctx->input(0);
    const Tensor& rt = ctx->input(1);
    double reg = 0.0;
    if (ctx->num_inputs() > 2 && ctx->input(2).NumElements() == 1)
      reg = (double)ctx->input(2).flat<double>()[0];
    BatchView bm, br;
    Status s1 = GetBatch(mt, &bm), s2 = GetBatch(rt, &br);
    if (!s1.ok() || !s2.ok() || br.m != bm.m || bm.nbatch != br.nbatch) {
      ctx->SetStatus(errors::InvalidArgument("solve_ls: bad shapes"));
      return;
    }
    int m = (int)bm.m, n = (int)bm.n, k = (int)br.n;
    TensorShape os;
    for (int i = 0; i < mt.shape().dims() - 2; ++i)
      os.AddDim(mt.shape().dim_size(i));
    os.AddDim(n);
    os.AddDim(k);
    Tensor* out = ctx->allocate_output(0, os);
    Mat a, ata((size_t)n * n);
    std::vector<int> piv;
    std::vector<double> col(n);
    for (int64_t b = 0; b < bm.nbatch; ++b) {
      LoadMat(mt.flat<T>() + b * m * n, a, (int64_t)m * n);
      // normal equations (A^T A + reg I) x = A^T b — the reference's fast path
      for (int i = 0; i < n; ++i)
        for (int j = 0; j < n; ++j) {
          double acc = i == j ? reg : 0.0;
          for (int l = 0; l < m; ++l) acc += a[l * n + i] * a[l * n + j];
          ata[i * n + j] = acc;
        }
      Mat lu = ata;
      if (LuFactor(lu, n, piv) == 0) {
        ctx->SetStatus(errors::InvalidArgument("solve_ls: singular A^T A"));
        return;
      }
      const T* rp = rt.flat<T>() + b * m * k;
      T* op = out->flat<T>() + b * n * k;
      for (int j = 0; j < k; ++j) {
        for (int i = 0; i < n; ++i) {
          double acc = 0.0;
          for (int l = 0; l < m; ++l) acc += a[l * n + i] * (double)rp[l * k + j];
          col[i] = acc;
        }
        LuSolveVec(lu, n, piv, col.data());
        for (int i = 0; i < n; ++i) op[i * k + j] = (T)col[i];
      }
    }
  }
};

#define REG_LINALG(NAME, OP)                                                   \
  REGISTER_KERNEL_BUILDER(                                                     \
      Name(NAME).Device(DEVICE_CPU).TypeConstraint<float>("T"), OP<float>);    \
  REGISTER_KERNEL_BUILDER(                                                     \
      Name(NAME).Device(DEVICE_CPU).TypeConstraint<double>("T"), OP<double>);

REG_LINALG("Cholesky", CholeskyOp)
REG_LINALG("CholeskyGrad", CholeskyGradOp)
REG_LINALG("MatrixDeterminant", MatrixDeterminantOp)
REG_LINALG("MatrixInverse", MatrixInverseOp)
REG_LINALG("MatrixSolve", MatrixSolveOp)
REG_LINALG("MatrixTriangularSolve", MatrixTriangularSolveOp)
REG_LINALG("MatrixSolveLs", MatrixSolveLsOp)
REG_LINALG("Qr", QrOp)
REG_LINALG("Svd", SvdOp)
REG_LINALG("SelfAdjointEigV2", SelfAdjointEigV2Op)
#undef REG_LINALG

}  // namespace
}  // namespace stf
