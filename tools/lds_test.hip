#include <hip/hip_runtime.h>
#include <cstdio>
__global__ void k(float* out, int n) {
  extern __shared__ __attribute__((aligned(16))) char sm[];
  float* f = (float*)sm;
  int tid = threadIdx.x;
  // touch the whole 68224-byte region
  for (int i = tid; i < 68224 / 4; i += blockDim.x) f[i] = (float)i;
  __syncthreads();
  if (tid < n) out[tid] = f[17000 - 1 + tid];  // near the top of the region
}
__global__ void kg(const float* src, float* out, int nbytes) {
  extern __shared__ __attribute__((aligned(16))) char sm[];
  // glds into high LDS offsets
  auto rsrc = __builtin_amdgcn_make_buffer_rsrc((void*)src, 0, nbytes, 0x00020000);
  int lane = threadIdx.x & 63;
  __builtin_amdgcn_raw_ptr_buffer_load_lds(
      rsrc, (__attribute__((address_space(3))) void*)&sm[67200], 16,
      (unsigned)(lane * 16), 0, 0, 0);
  __builtin_amdgcn_s_waitcnt(0);
  __syncthreads();
  float* f = (float*)&sm[67200];
  if (threadIdx.x < 64) out[threadIdx.x] = f[threadIdx.x];
}
int main() {
  float* out; hipMalloc(&out, 1024);
  float* src; hipMalloc(&src, 4096);
  float host[256]; for (int i = 0; i < 256; ++i) host[i] = i + 1;
  hipMemcpy(src, host, 1024, hipMemcpyHostToDevice);
  hipError_t e1 = hipFuncSetAttribute((const void*)k, hipFuncAttributeMaxDynamicSharedMemorySize, 68224);
  printf("setattr k: %s\n", hipGetErrorString(e1));
  hipLaunchKernelGGL(k, dim3(1), dim3(256), 68224, 0, out, 8);
  printf("launch k: %s\n", hipGetErrorString(hipGetLastError()));
  printf("sync k: %s\n", hipGetErrorString(hipDeviceSynchronize()));
  float h[64]; hipMemcpy(h, out, 64, hipMemcpyDeviceToHost);
  printf("k out: %.0f %.0f\n", h[0], h[1]);
  hipError_t e2 = hipFuncSetAttribute((const void*)kg, hipFuncAttributeMaxDynamicSharedMemorySize, 68224);
  printf("setattr kg: %s\n", hipGetErrorString(e2));
  hipLaunchKernelGGL(kg, dim3(1), dim3(64), 68224, 0, src, out, 1024);
  printf("launch kg: %s\n", hipGetErrorString(hipGetLastError()));
  printf("sync kg: %s\n", hipGetErrorString(hipDeviceSynchronize()));
  hipMemcpy(h, out, 256, hipMemcpyDeviceToHost);
  printf("kg out: %.0f %.0f %.0f (expect 1 2 3)\n", h[0], h[1], h[2]);
  return 0;
}
