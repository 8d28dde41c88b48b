// CDNA4 (gfx950) device kernels shared by the torch extension
// (k3samd_kernels.hip) and the standalone in-pod smoke binary
// (native/hipsmoke/mi_stream.hip). Pure HIP — no torch dependency.
//
// Design (MI355X-first):
//  * 16 B/lane (float4 ext-vector) loads/stores — 1 KiB per wave64 vector
//    memory instruction, the HBM3E coalescing sweet spot.
//  * non-temporal variants for the >L3 streaming regime (measured faster:
//    6.40 TB/s vs 5.83 TB/s plain triad on MI355X silicon).
//  * MFMA kernels use the gfx950 v_mfma_f32_16x16x32_bf16 shape; fragment
//    layout validated on silicon (lane l: A[l&15][(l>>4)*8+j]).

#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>

namespace k3samd_kern {

constexpr int kThreadsPerBlock = 256;  // 4 waves of 64

using f4 = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ f4 triad_op(const f4 b, const f4 c, float s) {
  return b + s * c;
}

template <bool NT>
__global__ void stream_triad_kernel(f4* __restrict__ a, const f4* __restrict__ b,
                                    const f4* __restrict__ c, float s,
                                    int64_t n4) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n4) return;
  if constexpr (NT) {
    f4 bv = __builtin_nontemporal_load(&b[i]);
    f4 cv = __builtin_nontemporal_load(&c[i]);
    __builtin_nontemporal_store(triad_op(bv, cv, s), &a[i]);
  } else {
    a[i] = triad_op(b[i], c[i], s);
  }
}

template <bool NT>
__global__ void stream_copy_kernel(f4* __restrict__ a, const f4* __restrict__ b,
                                   int64_t n4) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n4) return;
  if constexpr (NT) {
    __builtin_nontemporal_store(__builtin_nontemporal_load(&b[i]), &a[i]);
  } else {
    a[i] = b[i];
  }
}

template <bool NT>
__global__ void stream_scale_kernel(f4* __restrict__ a, const f4* __restrict__ c,
                                    float s, int64_t n4) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n4) return;
  if constexpr (NT) {
    __builtin_nontemporal_store(s * __builtin_nontemporal_load(&c[i]), &a[i]);
  } else {
    a[i] = s * c[i];
  }
}

template <bool NT>
__global__ void stream_add_kernel(f4* __restrict__ a, const f4* __restrict__ b,
                                  const f4* __restrict__ c, int64_t n4) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n4) return;
  if constexpr (NT) {
    f4 bv = __builtin_nontemporal_load(&b[i]);
    f4 cv = __builtin_nontemporal_load(&c[i]);
    __builtin_nontemporal_store(bv + cv, &a[i]);
  } else {
    a[i] = b[i] + c[i];
  }
}

// Grid-stride triad with U float4-elements per thread per pass — used by
// the tuning sweep (mi-stream --tune) to probe block-size/unroll space.
template <bool NT, int U>
__global__ void stream_triad_gs_kernel(f4* __restrict__ a,
                                       const f4* __restrict__ b,
                                       const f4* __restrict__ c, float s,
                                       int64_t n4) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i + (U - 1) * stride < n4; i += U * stride) {
    f4 bv[U], cv[U];
#pragma unroll
    for (int u = 0; u < U; ++u) {
      if constexpr (NT) {
        bv[u] = __builtin_nontemporal_load(&b[i + u * stride]);
        cv[u] = __builtin_nontemporal_load(&c[i + u * stride]);
      } else {
        bv[u] = b[i + u * stride];
        cv[u] = c[i + u * stride];
      }
    }
#pragma unroll
    for (int u = 0; u < U; ++u) {
      if constexpr (NT) {
        __builtin_nontemporal_store(triad_op(bv[u], cv[u], s),
                                    &a[i + u * stride]);
      } else {
        a[i + u * stride] = triad_op(bv[u], cv[u], s);
      }
    }
  }
  // tail
  for (; i < n4; i += stride) a[i] = triad_op(b[i], c[i], s);
}

// LDS-staged triad (SURVEY §2c's blueprint variant, kept for the measured
// comparison): stage b and c tiles into LDS with the async 16-byte
// `global_load_lds` DMA, then compute from LDS. For a pure streaming op
// with ZERO reuse this adds an LDS round trip per element, so the direct
// non-temporal kernel above is expected to win — mi-stream --lds-compare
// measures the gap rather than asserting it. The LDS destination of
// global_load_lds is wave-uniform base + lane*16 (per-lane scatter is
// not a thing), so each wave stages its own contiguous 64-f4 slice.
__global__ void stream_triad_lds_kernel(f4* __restrict__ a,
                                        const f4* __restrict__ b,
                                        const f4* __restrict__ c, float s,
                                        int64_t n4) {
#if defined(__gfx950__)
  __shared__ f4 sb[kThreadsPerBlock], sc[kThreadsPerBlock];
  const int tid = threadIdx.x;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + tid;
  // lanes past the end stage element 0 (the whole wave must issue the
  // DMA together); their results are simply not stored
  int64_t i_ld = i < n4 ? i : 0;
  f4* wave_sb = sb + (tid & ~63);  // wave-uniform LDS base
  f4* wave_sc = sc + (tid & ~63);
  __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)&b[i_ld],
                                   (__attribute__((address_space(3))) void*)wave_sb,
                                   16, 0, 0);
  __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) void*)&c[i_ld],
                                   (__attribute__((address_space(3))) void*)wave_sc,
                                   16, 0, 0);
  __syncthreads();  // drains the in-flight DMA (vmcnt(0) before barrier)
  if (i >= n4) return;
  __builtin_nontemporal_store(triad_op(sb[tid], sc[tid], s), &a[i]);
#else
  (void)a; (void)b; (void)c; (void)s; (void)n4;
#endif
}

// ---------------------------------------------------------------------------
// MFMA kernels (device code only selected when compiling for gfx950).
// ---------------------------------------------------------------------------

#if defined(__gfx950__)
#define K3_HAS_MFMA 1
using bf16x8 = __attribute__((ext_vector_type(8))) short;   // 8 bf16 = 4 VGPR
using f32x4 = __attribute__((ext_vector_type(4))) float;    // C/D for 16x16
using f32x16 = __attribute__((ext_vector_type(16))) float;  // C/D for 32x32
#else
#define K3_HAS_MFMA 0
#endif

// MFMA throughput: 4 independent accumulators per wave (> the 2 needed to
// reach the 32-cycle issue rate of v_mfma_f32_16x16x32_bf16, dependent
// latency 40 < 2x32). FLOPs per MFMA = 2*16*16*32 = 16384.
static __global__ void mfma_throughput_kernel(float* __restrict__ out,
                                              int iters) {
#if K3_HAS_MFMA
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = (short)(0x3f80 + ((threadIdx.x + j) & 7));
    b[j] = (short)(0x3f00 + ((threadIdx.x * 3 + j) & 7));
  }
  f32x4 acc0 = {0.f, 0.f, 0.f, 0.f}, acc1 = acc0, acc2 = acc0, acc3 = acc0;
  for (int i = 0; i < iters; ++i) {
    acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc0, 0, 0, 0);
    acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc1, 0, 0, 0);
    acc2 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc2, 0, 0, 0);
    acc3 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc3, 0, 0, 0);
  }
  float r = acc0[0] + acc1[1] + acc2[2] + acc3[3];
  if (threadIdx.x == 0) out[blockIdx.x] = r;  // keep the work alive
#else
  if (threadIdx.x == 0) out[blockIdx.x] = -1.f;
#endif
}

// 32x32 variant: v_mfma_f32_32x32x16_bf16 (64-cycle issue = dependent
// latency, so 2 independent accumulators saturate the pipe; the 32x32
// shape's measured ceiling is ~2382 TF vs ~2075 for 16x16).
// FLOPs per MFMA = 2*32*32*16 = 32768.
static __global__ void mfma_throughput32_kernel(float* __restrict__ out,
                                                int iters) {
#if K3_HAS_MFMA
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = (short)(0x3f80 + ((threadIdx.x + j) & 7));
    b[j] = (short)(0x3f00 + ((threadIdx.x * 3 + j) & 7));
  }
  // 4 chains x 2-deep unroll at high grid occupancy: the r02 sweep
  // (tools/mfma_tune.hip, gpurun_out/r2_mfma_tune.txt) measured 2477 TF
  // with ch=4 at 4096 blocks vs 2203 TF for the old 2-chain config —
  // 99 % of the 2.5 PF dense spec peak.
  f32x16 acc0 = {}, acc1 = {}, acc2 = {}, acc3 = {};
  for (int i = 0; i < iters; i += 2) {
    acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc0, 0, 0, 0);
    acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc1, 0, 0, 0);
    acc2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc2, 0, 0, 0);
    acc3 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc3, 0, 0, 0);
    acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc0, 0, 0, 0);
    acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc1, 0, 0, 0);
    acc2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc2, 0, 0, 0);
    acc3 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, acc3, 0, 0, 0);
  }
  float r = acc0[0] + acc1[1] + acc2[2] + acc3[3];
  if (threadIdx.x == 0) out[blockIdx.x] = r;
#else
  if (threadIdx.x == 0) out[blockIdx.x] = -1.f;
#endif
}

#if K3_HAS_MFMA
using i32x8 = __attribute__((ext_vector_type(8))) int;
#endif

// Block-scaled MX MFMA throughput (gfx950-only instruction class, the sole
// path to the chip's low-precision peaks): v_mfma_scale_f32_16x16x128_f8f6f4
// with FMT selecting the operand format (0 = fp8 e4m3, 4 = fp4). K = 128,
// so FLOPs per MFMA = 2*16*16*128 = 65536. Scales are E8M0 bias-127 (= 1.0);
// 4 independent accumulators as in the bf16 kernel.
template <int FMT>
static __global__ void mfma_throughput_mx_kernel(float* __restrict__ out,
                                                 int iters) {
#if K3_HAS_MFMA
  i32x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = (int)(0x3c3c3c3c + threadIdx.x + j);
    b[j] = (int)(0x35353535 + threadIdx.x * 3 + j);
  }
  const int scale = 0x7f7f7f7f;  // E8M0 1.0 in every byte
  f32x4 acc0 = {}, acc1 = {}, acc2 = {}, acc3 = {};
  for (int i = 0; i < iters; ++i) {
    acc0 = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc0, FMT, FMT, 0, scale, 0, scale);
    acc1 = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc1, FMT, FMT, 0, scale, 0, scale);
    acc2 = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc2, FMT, FMT, 0, scale, 0, scale);
    acc3 = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
        a, b, acc3, FMT, FMT, 0, scale, 0, scale);
  }
  float r = acc0[0] + acc1[1] + acc2[2] + acc3[3];
  if (threadIdx.x == 0) out[blockIdx.x] = r;
#else
  if (threadIdx.x == 0) out[blockIdx.x] = -1.f;
#endif
}

// Single-tile D[16x16] = A[16x32] * B[32x16] through one
// v_mfma_f32_16x16x32_bf16 for numerics validation.
//   layout 0 (validated on MI355X): lane l holds A[l&15][(l>>4)*8 + j]
//   layout 1 (rejected candidate) : A[l&15][(l>>4)*4 + (j&3) + 16*(j>>2)]
// B mirrors A with row/col swapped; C/D: col=lane&15, row=(lane>>4)*4+reg.
static __global__ void mfma_gemm16_kernel(const uint16_t* __restrict__ A,
                                          const uint16_t* __restrict__ B,
                                          float* __restrict__ D, int layout) {
#if K3_HAS_MFMA
  const int l = threadIdx.x;  // one wave
  bf16x8 af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int k = (layout == 0) ? ((l >> 4) * 8 + j)
                          : ((l >> 4) * 4 + (j & 3) + 16 * (j >> 2));
    af[j] = (short)A[(l & 15) * 32 + k];
    bf[j] = (short)B[k * 16 + (l & 15)];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    D[((l >> 4) * 4 + r) * 16 + (l & 15)] = acc[r];
  }
#else
  (void)A; (void)B; (void)D; (void)layout;
#endif
}

// Single-tile D[16x16] = dequant(A[16x128]) * dequant(B[128x16]) through
// one block-scaled v_mfma_scale_f32_16x16x128_f8f6f4 — numerics
// validation for the MX path (the throughput kernels above never check
// values). Scales fixed at E8M0 1.0 so dequant is the identity on the
// element encodings.
//
// Presumed A/B fragment layout (generalizing the silicon-validated bf16
// 16x16x32 map — 4 lanes per row, contiguous k-chunk per lane):
//   lane l holds A[l&15][(l>>4)*32 + j]  for j = 0..31 (fp8: byte j of
//   the 32-byte operand; fp4: nibble j of the 16-byte operand, low
//   nibble first), B mirrored with row/col swapped.
// C/D use the shape-determined map col=lane&15, row=(lane>>4)*4+reg
// (dtype-independent per the CDNA4 ISA).
//
// A is packed row-major (fp8: 1 B/elem, stride 128; fp4: 2 elems/B, low
// nibble = even k, stride 64 B). B is packed COLUMN-major with k
// contiguous (fp8: B_bytes[col*128 + k]; fp4: nibbles along k,
// B_bytes[col*64 + k/2]) so both gathers are contiguous per lane.
template <int FMT>  // 0 = fp8 e4m3, 4 = fp4 e2m1
static __global__ void mx_gemm16_kernel(const uint8_t* __restrict__ A,
                                        const uint8_t* __restrict__ B,
                                        float* __restrict__ D) {
#if K3_HAS_MFMA
  const int l = threadIdx.x;  // one wave
  i32x8 a = {0, 0, 0, 0, 0, 0, 0, 0}, b = {0, 0, 0, 0, 0, 0, 0, 0};
  uint8_t* ab = reinterpret_cast<uint8_t*>(&a);
  uint8_t* bb = reinterpret_cast<uint8_t*>(&b);
  if constexpr (FMT == 0) {
#pragma unroll
    for (int j = 0; j < 32; ++j) {
      int k = (l >> 4) * 32 + j;
      ab[j] = A[(l & 15) * 128 + k];
      bb[j] = B[(l & 15) * 128 + k];  // B packed col-major: col=l&15
    }
  } else {  // fp4: 32 nibbles = 16 bytes per lane
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      int kb = (l >> 4) * 16 + j;  // byte index along the packed k axis
      ab[j] = A[(l & 15) * 64 + kb];
      bb[j] = B[(l & 15) * 64 + kb];
    }
  }
  const int scale = 0x7f7f7f7f;  // E8M0 1.0
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
      a, b, acc, FMT, FMT, 0, scale, 0, scale);
  // Defensive: keep the scale VGPR alive past the MFMA. Without this,
  // LLVM allocates the destination v[0:3] OVER the scale source
  // (verified in the generated asm); MFMA sources are read across the
  // instruction's internal passes and the ISA's overlap rules for this
  // operand class are not documented — don't rely on it.
  __asm__ volatile("" ::"v"(scale));
#pragma unroll
  for (int r = 0; r < 4; ++r)
    D[((l >> 4) * 4 + r) * 16 + (l & 15)] = acc[r];
#else
  (void)A; (void)B; (void)D;
#endif
}

inline int64_t stream_grid(int64_t n4) {
  return (n4 + kThreadsPerBlock - 1) / kThreadsPerBlock;
}

// The AQL dispatch packet's grid_size_x is a 32-bit WORK-ITEM count, so a
// single flat launch caps at 2^32-1 threads — a 64 GiB fp32 buffer (2^32
// float4s) needs exactly 2^32 and fails with hipErrorInvalidConfiguration
// (hit in practice at buffer-MiB >= 65536; gpurun_out/r2_buffer_sweep.err).
// Launch in <= kMaxFlatChunk pieces instead of switching to grid-stride:
// the flat kernel beat every grid-stride tuning (profiles/
// r01_triad_tuning_sweep.txt) and a second launch per 64 GiB is free.
constexpr int64_t kMaxFlatChunk =
    (int64_t(4294967295u) / kThreadsPerBlock) * kThreadsPerBlock;

template <typename LaunchOne>
inline void launch_chunked(int64_t n4, LaunchOne&& launch_one) {
  if (n4 <= 0) return;  // a 0-block grid is an invalid launch
  int64_t off = 0;
  do {
    int64_t count = n4 - off;
    if (count > kMaxFlatChunk) count = kMaxFlatChunk;
    launch_one(off, count);
    off += count;
  } while (off < n4);
}

}  // namespace k3samd_kern
