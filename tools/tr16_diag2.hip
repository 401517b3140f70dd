#include <hip/hip_runtime.h>
#include <cstdio>
typedef __attribute__((__vector_size__(4 * sizeof(__bf16)))) __bf16 bf16x4_v;
#define LV_LDS __attribute__((address_space(3)))
__global__ void diag2(float* out) {
  __shared__ unsigned short panel[32 * 16];
  // Stage logical B[k][c] = k*16 + c into TILED [8][4][4][4] layout.
  for (int i = threadIdx.x; i < 512; i += 64) {
    const int row = i / 16, col = i % 16;
    float v = (float)(row * 16 + col);
    union { float f; unsigned int u; } cv; cv.f = v;
    panel[((row / 4) * 4 + col / 4) * 16 + (row % 4) * 4 + (col % 4)] =
        (unsigned short)(cv.u >> 16);
  }
  __syncthreads();
  const int lane = threadIdx.x;
  const int g = lane >> 4, cl = lane & 15;
  const int k0 = 0;
  const int kb = (k0 + g * 8) / 4;
  const int cb = cl / 4;
  const char* addr = reinterpret_cast<const char*>(panel) +
                     (((kb * 4 + cb) * 16) + (cl & 3)) * 2;
  bf16x4_v lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (LV_LDS bf16x4_v*)(addr));
  bf16x4_v hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
      (LV_LDS bf16x4_v*)(addr + 4 * 16 * 2));
  for (int j = 0; j < 4; ++j) {
    out[lane * 8 + j] = (float)lo[j];
    out[lane * 8 + 4 + j] = (float)hi[j];
  }
}
int main() {
  float* d; hipMalloc(&d, 64 * 8 * sizeof(float));
  hipLaunchKernelGGL(diag2, dim3(1), dim3(64), 0, 0, d);
  float h[512]; hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
  int bad = 0;
  for (int l = 0; l < 64; ++l) {
    const int g = l >> 4, cl = l & 15;
    for (int j = 0; j < 8; ++j) {
      float want = (float)((g * 8 + j) * 16 + cl);  // B[k0+g*8+j][cl]
      if (h[l * 8 + j] != want && bad < 12) {
        printf("lane %2d j %d got %5.0f want %5.0f\n", l, j,
               h[l * 8 + j], want);
        ++bad;
      }
    }
  }
  printf(bad ? "MISMATCHES above\n" : "ALL MATCH\n");
  return 0;
}
