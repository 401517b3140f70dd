#include <hip/hip_runtime.h>
#include <cstdio>
typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4_v;
#define LV_LDS __attribute__((address_space(3)))
__global__ void diag(float* out) {
  __shared__ unsigned short panel[32 * 16];
  // panel[k][c] = k*16 + c stored as bf16 value (exact)
  for (int i = threadIdx.x; i < 512; i += 64) {
    float v = (float)i;
    union { float f; unsigned int u; } cv; cv.f = v;
    panel[i] = (unsigned short)(cv.u >> 16);
  }
  __syncthreads();
  const int lane = threadIdx.x;
  // each lane passes addr = panel + lane*2 (element index = lane)
  const char* addr = reinterpret_cast<const char*>(panel) + lane * 2;
  bf16x4_v r = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (LV_LDS bf16x4_v*)(addr));
  for (int j = 0; j < 4; ++j) out[lane * 4 + j] = (float)r[j];
}
int main() {
  float* d; hipMalloc(&d, 64 * 4 * sizeof(float));
  hipLaunchKernelGGL(diag, dim3(1), dim3(64), 0, 0, d);
  float h[256]; hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  for (int l = 0; l < 64; ++l) {
    printf("lane %2d: %5.0f %5.0f %5.0f %5.0f\n", l,
           h[l*4], h[l*4+1], h[l*4+2], h[l*4+3]);
  }
  return 0;
}
