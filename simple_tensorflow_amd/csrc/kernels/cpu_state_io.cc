// Placeholder for SaveV2/RestoreV2 tensor-bundle kernels (filled in the
// checkpoint milestone; the Python Saver currently drives bundle IO).
#include "kernels/kernel_util.h"

namespace stf {}  // namespace stf
