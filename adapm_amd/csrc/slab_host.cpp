// Host-spill arena allocation: pinned, device-visible host memory.
// Separate TU so the pinned-alloc path is explicit HIP API (hipHostMalloc
// with hipHostMallocMapped) on GPU builds and plain calloc for CPU-only
// stores.
#include <hip/hip_runtime.h>

#include <cstdlib>
#include <cstring>
#include <stdexcept>

namespace adapm {
// mirror of the Slab members we need (defined in core.cpp); we implement
// its methods here via a forward declaration trick: core.cpp includes the
// full struct, so we just provide the out-of-line definitions.
}

// The actual definitions are provided via the functions below, called from
// core.cpp through C-style hooks to avoid duplicating the struct layout.
extern "C" {

int adapm_host_arena_alloc(long long floats, int want_device_visible, void** host_ptr,
                           void** dev_ptr) {
  size_t bytes = (size_t)floats * sizeof(float);
  if (want_device_visible) {
    void* p = nullptr;
    if (hipHostMalloc(&p, bytes, hipHostMallocMapped) != hipSuccess) return -1;
    std::memset(p, 0, bytes);
    void* d = nullptr;
    if (hipHostGetDevicePointer(&d, p, 0) != hipSuccess) {
      hipHostFree(p);
      return -2;
    }
    *host_ptr = p;
    *dev_ptr = d;
  } else {
    void* p = std::calloc(bytes, 1);
    if (!p) return -3;
    *host_ptr = p;
    *dev_ptr = p;
  }
  return 0;
}

void adapm_host_arena_free(void* host_ptr, int was_device_visible) {
  if (!host_ptr) return;
  if (was_device_visible)
    hipHostFree(host_ptr);
  else
    std::free(host_ptr);
}
}
