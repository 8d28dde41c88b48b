// Standalone probe of v_mfma_scale_f32_16x16x128_f8f6f4 semantics.
#include <cstdio>
#include <cstring>
#include <hip/hip_runtime.h>
using i32x8 = __attribute__((ext_vector_type(8))) int;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__global__ void probe(const uint8_t* A, const uint8_t* B, float* D,
                      int scale_word) {
  const int l = threadIdx.x;
  i32x8 a = {0,0,0,0,0,0,0,0}, b = {0,0,0,0,0,0,0,0};
  uint8_t* ab = reinterpret_cast<uint8_t*>(&a);
  uint8_t* bb = reinterpret_cast<uint8_t*>(&b);
  for (int j = 0; j < 32; ++j) {
    int k = (l >> 4) * 32 + j;
    ab[j] = A[(l & 15) * 128 + k];
    bb[j] = B[(l & 15) * 128 + k];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      a, b, acc, 0, 0, 0, scale_word, 0, scale_word);
  __asm__ volatile("" ::"v"(scale_word));
  for (int r = 0; r < 4; ++r)
    D[((l >> 4) * 4 + r) * 16 + (l & 15)] = acc[r];
}

int main() {
  uint8_t *A, *B; float* D;
  (void)hipMallocManaged(&A, 16 * 128);
  (void)hipMallocManaged(&B, 16 * 128);
  (void)hipMallocManaged(&D, 16 * 16 * 4);
  auto run = [&](const char* name, int scale_word) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, A, B, D, scale_word);
    hipError_t e = hipDeviceSynchronize();
    std::printf("%s (err=%s): D[0][0]=%g D[0][1]=%g D[1][0]=%g D[15][15]=%g\n",
                name, hipGetErrorString(e), (double)D[0], (double)D[1],
                (double)D[16], (double)D[255]);
  };
  // case 1: all zeros -> D must be 0
  memset(A, 0, 16 * 128); memset(B, 0, 16 * 128);
  run("zeros", 0x7f7f7f7f);
  // case 2: A[0][0]=1.0 (e4m3 0x38), B col0 k0 = 2.0 (0x40): D[0][0]=2
  memset(A, 0, 16 * 128); memset(B, 0, 16 * 128);
  A[0] = 0x38; B[0] = 0x40;
  run("a00=1,b00=2", 0x7f7f7f7f);
  // case 3: same data, scale word = plain 0x7f in byte 0 only
  run("scale=0x0000007f", 0x0000007f);
  // case 4: all A=1.0, all B=1.0 -> D[i][j] = 128
  memset(A, 0x38, 16 * 128); memset(B, 0x38, 16 * 128);
  run("all-ones", 0x7f7f7f7f);
  return 0;
}
