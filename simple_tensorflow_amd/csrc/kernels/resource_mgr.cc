// Minimal session-level resource manager (analog of the reference's
// ResourceMgr, core/framework/resource_mgr.h:103): queues and other shared
// stateful objects keyed by name. Kernels access it through
// OpKernelContext::resource_mgr.
#include "kernels/resource_mgr.h"

namespace stf {

void* NewResourceMgr() { return new ResourceMgr(); }
void DeleteResourceMgr(void* p) { delete static_cast<ResourceMgr*>(p); }

}  // namespace stf
