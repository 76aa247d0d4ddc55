#include <hip/hip_runtime.h>
#include <cstdio>
__global__ void addone(float* p, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] += 1.0f;
}
extern "C" int run_test() {
  float* d;
  hipError_t e = hipMalloc(&d, 1024 * 4);
  printf("malloc: %s\n", hipGetErrorString(e));
  hipMemset(d, 0, 1024 * 4);
  hipLaunchKernelGGL(addone, dim3(4), dim3(256), 0, 0, d, 1024);
  printf("launch: %s\n", hipGetErrorString(hipGetLastError()));
  e = hipDeviceSynchronize();
  printf("sync: %s\n", hipGetErrorString(e));
  float h[4];
  hipMemcpy(h, d, 16, hipMemcpyDeviceToHost);
  printf("val: %f\n", h[0]);
  return 0;
}
