#include "hip_util.h"

#include <hip/hip_runtime.h>

namespace xps {
namespace gpu {

#define XPS_HIP_CHECK(cmd)                                                            \
  do {                                                                                \
    hipError_t e_ = (cmd);                                                            \
    XPS_CHECK(e_ == hipSuccess) << "HIP error: " << hipGetErrorString(e_) << " in " #cmd; \
  } while (0)

int DeviceCount() {
  static int count = []() {
    int n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    if (e != hipSuccess) return 0;
    return n;
  }();
  return count;
}

bool Available() { return DeviceCount() > 0; }

SArray<char> StageToHost(const SArray<char>& dev) {
  XPS_CHECK(Available()) << "device SArray on a box with no GPU";
  SArray<char> host(dev.size());
  XPS_HIP_CHECK(hipMemcpy(host.data(), dev.data(), dev.size(), hipMemcpyDeviceToHost));
  return host;
}

void CopyHostToDevice(void* dst, const void* src, size_t n, int dev) {
  if (dev >= 0) XPS_HIP_CHECK(hipSetDevice(dev));
  XPS_HIP_CHECK(hipMemcpy(dst, src, n, hipMemcpyHostToDevice));
}

void CopyDeviceToHost(void* dst, const void* src, size_t n) {
  XPS_HIP_CHECK(hipMemcpy(dst, src, n, hipMemcpyDeviceToHost));
}

void DeviceSync(int dev) {
  if (dev >= 0) XPS_HIP_CHECK(hipSetDevice(dev));
  XPS_HIP_CHECK(hipDeviceSynchronize());
}

}  // namespace gpu
}  // namespace xps
