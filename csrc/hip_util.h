// Thin HIP runtime wrappers, runtime-gated so the library loads and the
// CPU paths run on machines without a GPU (the CI container).
#pragma once

#include <cstddef>
#include <cstdint>

#include "sarray.h"

namespace xps {
namespace gpu {

// number of HIP devices (0 on a GPU-less box; never throws)
int DeviceCount();
bool Available();

// synchronous copies for the slow/correctness TCP staging path
SArray<char> StageToHost(const SArray<char>& dev);          // D2H into fresh host buf
void CopyHostToDevice(void* dst, const void* src, size_t n, int dev);
void CopyDeviceToHost(void* dst, const void* src, size_t n);
void DeviceSync(int dev);

}  // namespace gpu
}  // namespace xps
