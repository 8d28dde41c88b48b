// k3samd_kernels.hip — CDNA4 (gfx950 / MI355X) native smoke & benchmark kernels.
//
// These kernels are the MI355X analog of the reference stack's in-pod GPU
// payload (K3S-NVidia runs `nvidia-smi` in its validation pod,
// /root/reference/nvidia-smi.yaml:13): a STREAM bandwidth suite that
// saturates HBM3E and an MFMA (matrix-core) warm-up/throughput kernel that
// lights up all 8 XCDs, with a single-tile MFMA GEMM for numerics checks.
//
// Design notes (MI355X-first, per CDNA4 guide):
//  * wave64: all block sizes are multiples of 64.
//  * STREAM kernels move 16 B/lane (float4) per instruction — the coalescing
//    sweet spot (64 lanes x 16 B = 1 KiB per vector instruction).
//  * grids are flat one-element-chunk-per-thread: a 1 GiB buffer yields
//    ~256Ki workgroups, vastly oversubscribing the 256 CUs across 8 XCDs.
//  * non-temporal variants (`__builtin_nontemporal_*`) bypass L2/LLC reuse
//    hints for the >L3 streaming regime; both variants are exposed so the
//    benchmark can pick the faster on real silicon.
//  * MFMA kernels target __builtin_amdgcn_mfma_f32_16x16x32_bf16 (gfx950
//    2xK shape) and only compile for gfx950.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#include <cstdint>

#define K3_CHECK(cond, msg)                                                    \
  TORCH_CHECK(cond, "k3samd: ", msg)

namespace {

constexpr int kThreadsPerBlock = 256;  // 4 waves of 64

// native clang vector type: 16 B loads/stores, componentwise arithmetic,
// and valid with the nontemporal builtins (HIP's float4 class is not)
using f4 = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ f4 triad_op(const f4 b, const f4 c, float s) {
  return b + s * c;
}

__device__ __forceinline__ f4 add_op(const f4 b, const f4 c) { return b + c; }

__device__ __forceinline__ f4 scale_op(const f4 c, float s) { return s * c; }

// ---------------------------------------------------------------------------
// STREAM kernels. Each thread owns one float4 (16 B). Plain and non-temporal
// flavors; the NT flavor streams past the caches (no reuse across STREAM
// iterations anyway once buffers exceed the 256 MiB Infinity Cache).
// ---------------------------------------------------------------------------

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
    __builtin_nontemporal_store(scale_op(__builtin_nontemporal_load(&c[i]), s),
                                &a[i]);
  } else {
    a[i] = scale_op(c[i], s);
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
    __builtin_nontemporal_store(add_op(bv, cv), &a[i]);
  } else {
    a[i] = add_op(b[i], c[i]);
  }
}

// ---------------------------------------------------------------------------
// MFMA kernels (gfx950 only).
// ---------------------------------------------------------------------------

#if defined(__gfx950__)
#define K3_HAS_MFMA 1
using bf16x8 = __attribute__((ext_vector_type(8))) short;   // 8 bf16 = 4 VGPR
using f32x4 = __attribute__((ext_vector_type(4))) float;    // C/D for 16x16
#else
#define K3_HAS_MFMA 0
#endif

// MFMA throughput: every wave hammers independent accumulators with
// v_mfma_f32_16x16x32_bf16 on register-resident fragments. 4 independent
// accumulators per wave > the 2 needed to reach the 32-cycle issue rate
// (dependent latency 40 < 2x32). FLOPs per MFMA = 2*16*16*32 = 16384.
__global__ void mfma_throughput_kernel(float* __restrict__ out, int iters) {
#if K3_HAS_MFMA
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    // arbitrary small bf16 patterns; value content is irrelevant for rate
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

// Single-tile GEMM D[16x16] = A[16x32] * B[32x16] through one
// mfma_f32_16x16x32_bf16, for numerics validation against torch fp32.
//
// Fragment layouts are not fully documented at HIP level; LAYOUT selects the
// candidate A/B lane->element mapping (validated empirically on gfx950, see
// tests/test_mfma_gpu.py):
//   layout 0: lane l holds A[l&15][ (l>>4)*8 + j ]            j=0..7
//   layout 1: lane l holds A[l&15][ (l>>4)*4 + (j&3) + 16*(j>>2) ]
// B mirrors A with row/col swapped: B[k][l&15] at the same k mapping.
// C/D (per guide): col = lane&15, row = (lane>>4)*4 + reg.
__global__ void mfma_gemm16_kernel(const uint16_t* __restrict__ A,
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
    int row = (l >> 4) * 4 + r;
    int col = l & 15;
    D[row * 16 + col] = acc[r];
  }
#endif
}

inline int64_t grid_for(int64_t n4) {
  return (n4 + kThreadsPerBlock - 1) / kThreadsPerBlock;
}

void check_stream_args(const torch::Tensor& t) {
  K3_CHECK(t.is_cuda(), "tensor must be on GPU");
  K3_CHECK(t.scalar_type() == torch::kFloat32, "tensor must be float32");
  K3_CHECK(t.is_contiguous(), "tensor must be contiguous");
  K3_CHECK(t.numel() % 4 == 0, "numel must be divisible by 4");
}

}  // namespace

// --------------------------- host entry points -----------------------------

void stream_triad(torch::Tensor a, torch::Tensor b, torch::Tensor c, double s,
                  bool nontemporal) {
  check_stream_args(a);
  check_stream_args(b);
  check_stream_args(c);
  K3_CHECK(a.numel() == b.numel() && a.numel() == c.numel(), "size mismatch");
  int64_t n4 = a.numel() / 4;
  auto stream = at::hip::getCurrentHIPStream();
  auto* ap = reinterpret_cast<f4*>(a.data_ptr<float>());
  auto* bp = reinterpret_cast<const f4*>(b.data_ptr<float>());
  auto* cp = reinterpret_cast<const f4*>(c.data_ptr<float>());
  if (nontemporal) {
    hipLaunchKernelGGL(stream_triad_kernel<true>, dim3(grid_for(n4)),
                       dim3(kThreadsPerBlock), 0, stream, ap, bp, cp, (float)s,
                       n4);
  } else {
    hipLaunchKernelGGL(stream_triad_kernel<false>, dim3(grid_for(n4)),
                       dim3(kThreadsPerBlock), 0, stream, ap, bp, cp, (float)s,
                       n4);
  }
  C10_HIP_KERNEL_LAUNCH_CHECK();
}

void stream_copy(torch::Tensor a, torch::Tensor b, bool nontemporal) {
  check_stream_args(a);
  check_stream_args(b);
  K3_CHECK(a.numel() == b.numel(), "size mismatch");
  int64_t n4 = a.numel() / 4;
  auto stream = at::hip::getCurrentHIPStream();
  auto* ap = reinterpret_cast<f4*>(a.data_ptr<float>());
  auto* bp = reinterpret_cast<const f4*>(b.data_ptr<float>());
  if (nontemporal) {
    hipLaunchKernelGGL(stream_copy_kernel<true>, dim3(grid_for(n4)),
                       dim3(kThreadsPerBlock), 0, stream, ap, bp, n4);
  } else {
    hipLaunchKernelGGL(stream_copy_kernel<false>, dim3(grid_for(n4)),
                       dim3(kThreadsPerBlock), 0, stream, ap, bp, n4);
  }
  C10_HIP_KERNEL_LAUNCH_CHECK();
}

void stream_scale(torch::Tensor a, torch::Tensor c, double s, bool nontemporal) {
  check_stream_args(a);
  check_stream_args(c);
  K3_CHECK(a.numel() == c.numel(), "size mismatch");
  int64_t n4 = a.numel() / 4;
  auto stream = at::hip::getCurrentHIPStream();
  auto* ap = reinterpret_cast<f4*>(a.data_ptr<float>());
  auto* cp = reinterpret_cast<const f4*>(c.data_ptr<float>());
  if (nontemporal) {
    hipLaunchKernelGGL(stream_scale_kernel<true>, dim3(grid_for(n4)),
                       dim3(kThreadsPerBlock), 0, stream, ap, cp, (float)s, n4);
  } else {
    hipLaunchKernelGGL(stream_scale_kernel<false>, dim3(grid_for(n4)),
                       dim3(kThreadsPerBlock), 0, stream, ap, cp, (float)s, n4);
  }
  C10_HIP_KERNEL_LAUNCH_CHECK();
}

void stream_add(torch::Tensor a, torch::Tensor b, torch::Tensor c,
                bool nontemporal) {
  check_stream_args(a);
  check_stream_args(b);
  check_stream_args(c);
  K3_CHECK(a.numel() == b.numel() && a.numel() == c.numel(), "size mismatch");
  int64_t n4 = a.numel() / 4;
  auto stream = at::hip::getCurrentHIPStream();
  auto* ap = reinterpret_cast<f4*>(a.data_ptr<float>());
  auto* bp = reinterpret_cast<const f4*>(b.data_ptr<float>());
  auto* cp = reinterpret_cast<const f4*>(c.data_ptr<float>());
  if (nontemporal) {
    hipLaunchKernelGGL(stream_add_kernel<true>, dim3(grid_for(n4)),
                       dim3(kThreadsPerBlock), 0, stream, ap, bp, cp, n4);
  } else {
    hipLaunchKernelGGL(stream_add_kernel<false>, dim3(grid_for(n4)),
                       dim3(kThreadsPerBlock), 0, stream, ap, bp, cp, n4);
  }
  C10_HIP_KERNEL_LAUNCH_CHECK();
}

// Launch `blocks` blocks x 256 threads, `iters` MFMA quads each.
// Returns total FLOPs issued so the caller can compute TFLOP/s.
double mfma_throughput(torch::Tensor out, int64_t iters) {
  K3_CHECK(out.is_cuda() && out.scalar_type() == torch::kFloat32 &&
               out.is_contiguous(),
           "out must be contiguous float32 GPU tensor");
  int64_t blocks = out.numel();
  K3_CHECK(blocks > 0 && blocks <= (1 << 22), "bad block count");
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_throughput_kernel, dim3(blocks),
                     dim3(kThreadsPerBlock), 0, stream, out.data_ptr<float>(),
                     (int)iters);
  C10_HIP_KERNEL_LAUNCH_CHECK();
  // waves/block = 256/64 = 4; 4 accumulators; 16384 FLOP per MFMA
  return (double)blocks * 4.0 * 4.0 * 16384.0 * (double)iters;
}

torch::Tensor mfma_gemm16(torch::Tensor A, torch::Tensor B, int64_t layout) {
  K3_CHECK(A.is_cuda() && B.is_cuda(), "A,B must be on GPU");
  K3_CHECK(A.scalar_type() == torch::kBFloat16 &&
               B.scalar_type() == torch::kBFloat16,
           "A,B must be bf16");
  K3_CHECK(A.is_contiguous() && B.is_contiguous(), "A,B must be contiguous");
  K3_CHECK(A.size(0) == 16 && A.size(1) == 32 && B.size(0) == 32 &&
               B.size(1) == 16,
           "A must be [16,32], B [32,16]");
  auto D = torch::empty({16, 16}, A.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(mfma_gemm16_kernel, dim3(1), dim3(64), 0, stream,
                     reinterpret_cast<const uint16_t*>(A.data_ptr()),
                     reinterpret_cast<const uint16_t*>(B.data_ptr()),
                     D.data_ptr<float>(), (int)layout);
  C10_HIP_KERNEL_LAUNCH_CHECK();
  return D;
}

// Host-side query: K3_HAS_MFMA is a device-compilation-pass macro, so check
// the actual device architecture at runtime instead.
bool has_mfma() {
  int dev = 0;
  if (hipGetDevice(&dev) != hipSuccess) return false;
  hipDeviceProp_t prop;
  if (hipGetDeviceProperties(&prop, dev) != hipSuccess) return false;
  return std::string(prop.gcnArchName).find("gfx950") != std::string::npos;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("stream_triad", &stream_triad, "STREAM triad a = b + s*c (fp32)",
        py::arg("a"), py::arg("b"), py::arg("c"), py::arg("s"),
        py::arg("nontemporal") = false);
  m.def("stream_copy", &stream_copy, py::arg("a"), py::arg("b"),
        py::arg("nontemporal") = false);
  m.def("stream_scale", &stream_scale, py::arg("a"), py::arg("c"), py::arg("s"),
        py::arg("nontemporal") = false);
  m.def("stream_add", &stream_add, py::arg("a"), py::arg("b"), py::arg("c"),
        py::arg("nontemporal") = false);
  m.def("mfma_throughput", &mfma_throughput, py::arg("out"), py::arg("iters"));
  m.def("mfma_gemm16", &mfma_gemm16, py::arg("A"), py::arg("B"),
        py::arg("layout") = 0);
  m.def("has_mfma", &has_mfma);
}
