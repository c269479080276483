// Self-test for hipGraph stream capture under the executor's usage pattern
// (multi-thread enqueue to one stream, hipMalloc during capture).
#include <hip/hip_runtime.h>

#include <sstream>
#include <thread>

#include "core/base.h"

namespace stf {

std::string HipGraphSelfTest() {
  std::ostringstream out;
  auto ck = [&](const char* what, hipError_t e) {
    out << what << "=" << hipGetErrorString(e) << "; ";
    return e == hipSuccess;
  };
  hipStream_t s;
  ck("create", hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
  void* buf = nullptr;
  ck("malloc", hipMalloc(&buf, 4096));
  ck("begin", hipStreamBeginCapture(s, hipStreamCaptureModeRelaxed));
  ck("memset_same_thread", hipMemsetAsync(buf, 1, 4096, s));
  // cross-thread enqueue
  hipError_t te = hipSuccess;
  std::thread t([&]() { te = hipMemsetAsync(buf, 2, 4096, s); });
  t.join();
  ck("memset_other_thread", te);
  // hipMalloc during capture (BFC growth case)
  void* buf2 = nullptr;
  ck("malloc_during_capture", hipMalloc(&buf2, 4096));
  ck("memset3", hipMemsetAsync(buf2, 3, 4096, s));
  hipGraph_t g = nullptr;
  ck("end", hipStreamEndCapture(s, &g));
  hipGraphExec_t exec = nullptr;
  if (g) {
    ck("instantiate", hipGraphInstantiate(&exec, g, nullptr, nullptr, 0));
    hipGraphDestroy(g);
  }
  if (exec) {
    ck("launch", hipGraphLaunch(exec, s));
    ck("sync", hipStreamSynchronize(s));
    unsigned char host[1];
    hipMemcpy(host, buf2, 1, hipMemcpyDeviceToHost);
    out << "val=" << (int)host[0] << "; ";
    hipGraphExecDestroy(exec);
  }
  hipFree(buf);
  if (buf2) hipFree(buf2);
  hipStreamDestroy(s);
  return out.str();
}

}  // namespace stf
