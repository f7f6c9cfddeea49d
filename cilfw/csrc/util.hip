// cilfw — small runtime utilities exposed to the ctypes wrapper.
#include <hip/hip_runtime.h>

extern "C" {

// drain the device and report the first sticky error (debug aid; the wrapper
// calls this under CILFW_SYNC_DEBUG=1 after every launch)
int cilfw_sync() {
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) return (int)e;
  return (int)hipDeviceSynchronize();
}

const char* cilfw_error_string(int e) { return hipGetErrorString((hipError_t)e); }

int cilfw_device_count() {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n;
}

}  // extern "C"
