// mfma_tune — occupancy/unroll sweep for the bf16 MFMA throughput
// kernels. The ladder's 32x32x16 number (~2203 TF) sits ~7-12 % below
// the chip's measured µbench floor (~2382 TF) and tuned peak (~2495 TF);
// the sweep probes the two levers that matter for a dependency-free MFMA
// stream: waves per SIMD (partner-wave MFMAs interleave in the shared
// pipe — more waves is NOT free) and accumulator chains per wave.
//
// Build: hipcc --offload-arch=gfx950 -O3 tools/mfma_tune.hip -o tools/mfma-tune
#include <cstdio>
#include <hip/hip_runtime.h>

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x16 = __attribute__((ext_vector_type(16))) float;

// CH = independent accumulator chains, U = unroll depth per chain pass
template <int CH, int U>
__global__ void mfma32_kernel(float* __restrict__ out, int iters) {
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = (short)(0x3f80 + ((threadIdx.x + j) & 7));
    b[j] = (short)(0x3f00 + ((threadIdx.x * 3 + j) & 7));
  }
  f32x16 acc[CH] = {};
  for (int i = 0; i < iters; i += U) {
#pragma unroll
    for (int u = 0; u < U; ++u)
#pragma unroll
      for (int ch = 0; ch < CH; ++ch)
        acc[ch] =
            __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc[ch], 0, 0, 0);
  }
  float r = 0;
#pragma unroll
  for (int ch = 0; ch < CH; ++ch) r += acc[ch][ch & 15];
  if (threadIdx.x == 0) out[blockIdx.x] = r;
}

using i32x8 = __attribute__((ext_vector_type(8))) int;

// Block-scaled MX, 32x32x64 shape (2*32*32*64 = 131072 FLOP per instr —
// same FLOPs as 16x16x128 but half the instruction count per FLOP at
// 32x32's issue pattern). FMT: 0=fp8 e4m3, 2=fp6, 4=fp4.
template <int FMT, int CH>
__global__ void mx32_kernel(float* __restrict__ out, int iters) {
  i32x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = (int)(0x3c3c3c3c + threadIdx.x + j);
    b[j] = (int)(0x35353535 + threadIdx.x * 3 + j);
  }
  const int scale = 0x7f7f7f7f;
  f32x16 acc[CH] = {};
  for (int i = 0; i < iters; ++i) {
#pragma unroll
    for (int ch = 0; ch < CH; ++ch)
      acc[ch] = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(
          a, b, acc[ch], FMT, FMT, 0, scale, 0, scale);
  }
  float r = 0;
#pragma unroll
  for (int ch = 0; ch < CH; ++ch) r += acc[ch][ch & 15];
  if (threadIdx.x == 0) out[blockIdx.x] = r;
}

int main(int argc, char** argv) {
  int iters = argc > 1 ? atoi(argv[1]) : 4096;
  float* out;
  (void)hipMalloc(&out, 65536 * sizeof(float));
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);

  auto bench = [&](auto kern, int blocks, int tpb, int ch,
                   const char* tag) {
    // one warm launch, then best-of-3
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(tpb), 0, 0, out, 256);
    (void)hipDeviceSynchronize();
    float best = 1e30f;
    for (int rep = 0; rep < 3; ++rep) {
      (void)hipEventRecord(e0);
      hipLaunchKernelGGL(kern, dim3(blocks), dim3(tpb), 0, 0, out, iters);
      (void)hipEventRecord(e1);
      (void)hipEventSynchronize(e1);
      float ms;
      (void)hipEventElapsedTime(&ms, e0, e1);
      if (ms < best) best = ms;
    }
    // FLOPs: blocks * (tpb/64) waves * ch chains * iters * 2*32*32*16
    double flops = (double)blocks * (tpb / 64) * ch * iters * 32768.0;
    std::printf("%-18s blocks=%5d tpb=%4d waves/CU=%4.1f  %8.1f TF\n", tag,
                blocks, tpb, blocks * (tpb / 64.0) / 256.0,
                flops / best / 1e9);
  };

  std::printf("32x32x16 bf16 MFMA sweep (iters=%d)\n", iters);
  for (int blocks : {256, 512, 1024, 2048, 4096}) {
    bench(mfma32_kernel<1, 4>, blocks, 256, 1, "ch=1 u=4");
    bench(mfma32_kernel<2, 2>, blocks, 256, 2, "ch=2 u=2");
    bench(mfma32_kernel<4, 2>, blocks, 256, 4, "ch=4 u=2");
  }
  // single wave per block (occupancy floor): 1 wave/SIMD at 1024 blocks
  for (int blocks : {1024, 2048, 4096}) {
    bench(mfma32_kernel<2, 2>, blocks, 64, 2, "ch=2 u=2 w64");
    bench(mfma32_kernel<4, 2>, blocks, 64, 4, "ch=4 u=2 w64");
  }

  // MX 32x32x64 vs the ladder's 16x16x128 (both 131072/65536 FLOP/instr)
  auto bench_mx = [&](auto kern, int blocks, int ch, double fpi,
                      const char* tag) {
    hipLaunchKernelGGL(kern, dim3(blocks), dim3(256), 0, 0, out, 256);
    (void)hipDeviceSynchronize();
    float best = 1e30f;
    for (int rep = 0; rep < 3; ++rep) {
      (void)hipEventRecord(e0);
      hipLaunchKernelGGL(kern, dim3(blocks), dim3(256), 0, 0, out, iters);
      (void)hipEventRecord(e1);
      (void)hipEventSynchronize(e1);
      float ms;
      (void)hipEventElapsedTime(&ms, e0, e1);
      if (ms < best) best = ms;
    }
    double flops = (double)blocks * 4 * ch * iters * fpi;
    std::printf("%-18s blocks=%5d  %8.1f TF\n", tag, blocks,
                flops / best / 1e9);
  };
  std::printf("MX 32x32x64 sweep\n");
  for (int blocks : {2048, 4096}) {
    bench_mx(mx32_kernel<0, 2>, blocks, 2, 131072.0, "mx32 fp8 ch=2");
    bench_mx(mx32_kernel<0, 4>, blocks, 4, 131072.0, "mx32 fp8 ch=4");
    bench_mx(mx32_kernel<2, 2>, blocks, 2, 131072.0, "mx32 fp6 ch=2");
    bench_mx(mx32_kernel<4, 2>, blocks, 2, 131072.0, "mx32 fp4 ch=2");
    bench_mx(mx32_kernel<4, 4>, blocks, 4, 131072.0, "mx32 fp4 ch=4");
  }
  return 0;
}
