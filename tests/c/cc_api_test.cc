// C++ API smoke: Scope graph build, ClientSession run, AddSymbolicGradients
// (reference cc/framework/gradients_test.cc / client_session_test.cc analog).
#include <cmath>
#include <cstdio>

#include "cc/cc_api.h"

using namespace stf;
using namespace stf::cc;

#define CHECK_TRUE(cond, msg)                         \
  if (!(cond)) {                                      \
    fprintf(stderr, "FAIL: %s\n", msg);               \
    return 1;                                         \
  }

int main() {
  Scope root = Scope::NewRootScope();
  // y = relu(x*w)^2, scalar-ish matmul 1x2 * 2x1
  Output x = ops::Const(root, {1.0f, -2.0f}, {1, 2});
  Output w = ops::Const(root, {3.0f, 0.5f}, {2, 1});
  Output xw = ops::MatMul(root, x, w);          // 1*3 + (-2)*0.5 = 2
  Output y = ops::Square(root, ops::Relu(root, xw));  // 4

  ClientSession session(root);
  std::vector<Tensor> out;
  Status s = session.Run({y}, &out);
  CHECK_TRUE(s.ok(), s.ToString().c_str());
  CHECK_TRUE(std::abs(out[0].flat<float>()[0] - 4.0f) < 1e-5, "y value");

  // dy/dw = 2*relu(xw)*x^T = 2*2*[1,-2]^T = [4,-8]
  std::vector<Output> grads;
  s = AddSymbolicGradients(root, {y}, {w}, &grads);
  CHECK_TRUE(s.ok(), s.ToString().c_str());
  s = session.Run({grads[0]}, &out);
  CHECK_TRUE(s.ok(), s.ToString().c_str());
  const float* g = out[0].flat<float>();
  CHECK_TRUE(std::abs(g[0] - 4.0f) < 1e-5, "dw[0]");
  CHECK_TRUE(std::abs(g[1] + 8.0f) < 1e-5, "dw[1]");

  // mean-loss gradient with accumulation: z = mean(x*a + x*a)
  Scope s2 = Scope::NewRootScope();
  Output a = ops::Const(s2, {1.0f, 2.0f, 3.0f, 4.0f}, {4});
  Output b = ops::Add(s2, ops::Mul(s2, a, a), ops::Mul(s2, a, a));
  Output z = ops::ReduceMean(s2, b, {0});
  std::vector<Output> g2;
  Status st = AddSymbolicGradients(s2, {z}, {a}, &g2);
  CHECK_TRUE(st.ok(), st.ToString().c_str());
  ClientSession sess2(s2);
  st = sess2.Run({g2[0]}, &out);
  CHECK_TRUE(st.ok(), st.ToString().c_str());
  // d/da mean(2a^2) = 4a/4 = a
  for (int i = 0; i < 4; ++i)
    CHECK_TRUE(std::abs(out[0].flat<float>()[i] - (float)(i + 1)) < 1e-5,
               "mean grad");

  printf("CC_API_OK\n");
  return 0;
}
