// k3samd_kernels.hip — torch bindings for the CDNA4 smoke/bench kernels.
//
// Device code lives in stream_kernels.h (shared with the standalone in-pod
// binary native/hipsmoke/mi_stream.hip). These kernels are the MI355X
// analog of the reference stack's in-pod GPU payload (K3S-NVidia runs
// `nvidia-smi` in its validation pod, /root/reference/nvidia-smi.yaml:13).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "stream_kernels.h"

namespace {

using k3samd_kern::f4;
using k3samd_kern::kThreadsPerBlock;
using k3samd_kern::stream_grid;

#define K3_CHECK(cond, msg) TORCH_CHECK(cond, "k3samd: ", msg)

void check_stream_args(const torch::Tensor& t) {
  K3_CHECK(t.is_cuda(), "tensor must be on GPU");
  K3_CHECK(t.scalar_type() == torch::kFloat32, "tensor must be float32");
  K3_CHECK(t.is_contiguous(), "tensor must be contiguous");
  K3_CHECK(t.numel() % 4 == 0, "numel must be divisible by 4");
}

}  // namespace

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
  k3samd_kern::launch_chunked(n4, [&](int64_t off, int64_t cnt) {
    if (nontemporal) {
      hipLaunchKernelGGL(k3samd_kern::stream_triad_kernel<true>,
                         dim3(stream_grid(cnt)), dim3(kThreadsPerBlock), 0,
                         stream, ap + off, bp + off, cp + off, (float)s, cnt);
    } else {
      hipLaunchKernelGGL(k3samd_kern::stream_triad_kernel<false>,
                         dim3(stream_grid(cnt)), dim3(kThreadsPerBlock), 0,
                         stream, ap + off, bp + off, cp + off, (float)s, cnt);
    }
  });
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
  k3samd_kern::launch_chunked(n4, [&](int64_t off, int64_t cnt) {
    if (nontemporal) {
      hipLaunchKernelGGL(k3samd_kern::stream_copy_kernel<true>,
                         dim3(stream_grid(cnt)), dim3(kThreadsPerBlock), 0,
                         stream, ap + off, bp + off, cnt);
    } else {
      hipLaunchKernelGGL(k3samd_kern::stream_copy_kernel<false>,
                         dim3(stream_grid(cnt)), dim3(kThreadsPerBlock), 0,
                         stream, ap + off, bp + off, cnt);
    }
  });
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
  k3samd_kern::launch_chunked(n4, [&](int64_t off, int64_t cnt) {
    if (nontemporal) {
      hipLaunchKernelGGL(k3samd_kern::stream_scale_kernel<true>,
                         dim3(stream_grid(cnt)), dim3(kThreadsPerBlock), 0,
                         stream, ap + off, cp + off, (float)s, cnt);
    } else {
      hipLaunchKernelGGL(k3samd_kern::stream_scale_kernel<false>,
                         dim3(stream_grid(cnt)), dim3(kThreadsPerBlock), 0,
                         stream, ap + off, cp + off, (float)s, cnt);
    }
  });
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
  k3samd_kern::launch_chunked(n4, [&](int64_t off, int64_t cnt) {
    if (nontemporal) {
      hipLaunchKernelGGL(k3samd_kern::stream_add_kernel<true>,
                         dim3(stream_grid(cnt)), dim3(kThreadsPerBlock), 0,
                         stream, ap + off, bp + off, cp + off, cnt);
    } else {
      hipLaunchKernelGGL(k3samd_kern::stream_add_kernel<false>,
                         dim3(stream_grid(cnt)), dim3(kThreadsPerBlock), 0,
                         stream, ap + off, bp + off, cp + off, cnt);
    }
  });
  C10_HIP_KERNEL_LAUNCH_CHECK();
}

// Launch `out.numel()` blocks x 256 threads, `iters` MFMA quads each.
// Returns total FLOPs issued so the caller can compute TFLOP/s.
double mfma_throughput(torch::Tensor out, int64_t iters, int64_t shape) {
  K3_CHECK(out.is_cuda() && out.scalar_type() == torch::kFloat32 &&
               out.is_contiguous(),
           "out must be contiguous float32 GPU tensor");
  K3_CHECK(shape == 16 || shape == 32 || shape == 8 || shape == 4,
           "shape must be 16, 32 (bf16), 8 (MX-fp8) or 4 (MX-fp4)");
  int64_t blocks = out.numel();
  K3_CHECK(blocks > 0 && blocks <= (1 << 22), "bad block count");
  auto stream = at::hip::getCurrentHIPStream();
  // shape 16: 4 chains x 16384 FLOP per iter; shape 32: 4 chains x 2
  // unroll per i+=2 step = 4 MFMAs/iter x 32768 FLOP
  double flops_per_wave_iter = shape == 32 ? 4.0 * 32768.0 : 4.0 * 16384.0;
  if (shape == 16) {
    hipLaunchKernelGGL(k3samd_kern::mfma_throughput_kernel, dim3(blocks),
                       dim3(kThreadsPerBlock), 0, stream,
                       out.data_ptr<float>(), (int)iters);
  } else if (shape == 32) {
    hipLaunchKernelGGL(k3samd_kern::mfma_throughput32_kernel, dim3(blocks),
                       dim3(kThreadsPerBlock), 0, stream,
                       out.data_ptr<float>(), (int)iters);
  } else {
    flops_per_wave_iter = 4.0 * 65536.0;  // K=128 MX shapes
    if (shape == 8) {
      hipLaunchKernelGGL(k3samd_kern::mfma_throughput_mx_kernel<0>,
                         dim3(blocks), dim3(kThreadsPerBlock), 0, stream,
                         out.data_ptr<float>(), (int)iters);
    } else {
      hipLaunchKernelGGL(k3samd_kern::mfma_throughput_mx_kernel<4>,
                         dim3(blocks), dim3(kThreadsPerBlock), 0, stream,
                         out.data_ptr<float>(), (int)iters);
    }
  }
  C10_HIP_KERNEL_LAUNCH_CHECK();
  return (double)blocks * 4.0 * flops_per_wave_iter * (double)iters;
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
  hipLaunchKernelGGL(k3samd_kern::mfma_gemm16_kernel, dim3(1), dim3(64), 0,
                     stream,
                     reinterpret_cast<const uint16_t*>(A.data_ptr()),
                     reinterpret_cast<const uint16_t*>(B.data_ptr()),
                     D.data_ptr<float>(), (int)layout);
  C10_HIP_KERNEL_LAUNCH_CHECK();
  return D;
}

// Block-scaled MX single-tile GEMM: D[16,16] = A[16,128] x B[128,16]
// with fp8-e4m3 (fmt=0) or fp4-e2m1 (fmt=4) operands, unit scales.
// A: uint8 [16,128] (fp8) / [16,64] (fp4, 2 elems per byte, low nibble =
// even k). B: uint8 packed COLUMN-major with k contiguous: [16,128]
// (fp8: row c holds column c of B) / [16,64] (fp4).
torch::Tensor mx_gemm16(torch::Tensor A, torch::Tensor B, int64_t fmt) {
  K3_CHECK(A.is_cuda() && B.is_cuda(), "A,B must be on GPU");
  K3_CHECK(A.scalar_type() == torch::kUInt8 &&
               B.scalar_type() == torch::kUInt8,
           "A,B must be uint8 (packed fp8/fp4)");
  K3_CHECK(A.is_contiguous() && B.is_contiguous(), "A,B must be contiguous");
  K3_CHECK(fmt == 0 || fmt == 4, "fmt must be 0 (fp8 e4m3) or 4 (fp4)");
  int64_t kb = fmt == 0 ? 128 : 64;  // packed bytes along k
  K3_CHECK(A.size(0) == 16 && A.size(1) == kb && B.size(0) == 16 &&
               B.size(1) == kb,
           "A must be [16,KB], B [16,KB] (B column-major-packed)");
  auto D = torch::empty({16, 16}, A.options().dtype(torch::kFloat32));
  auto stream = at::hip::getCurrentHIPStream();
  if (fmt == 0) {
    hipLaunchKernelGGL(k3samd_kern::mx_gemm16_kernel<0>, dim3(1), dim3(64),
                       0, stream,
                       reinterpret_cast<const uint8_t*>(A.data_ptr()),
                       reinterpret_cast<const uint8_t*>(B.data_ptr()),
                       D.data_ptr<float>());
  } else {
    hipLaunchKernelGGL(k3samd_kern::mx_gemm16_kernel<4>, dim3(1), dim3(64),
                       0, stream,
                       reinterpret_cast<const uint8_t*>(A.data_ptr()),
                       reinterpret_cast<const uint8_t*>(B.data_ptr()),
                       D.data_ptr<float>());
  }
  C10_HIP_KERNEL_LAUNCH_CHECK();
  return D;
}

// Host-side query: the extension only ships gfx950 code objects, so report
// whether the active device is gfx950.
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
  m.def("stream_scale", &stream_scale, py::arg("a"), py::arg("c"),
        py::arg("s"), py::arg("nontemporal") = false);
  m.def("stream_add", &stream_add, py::arg("a"), py::arg("b"), py::arg("c"),
        py::arg("nontemporal") = false);
  m.def("mfma_throughput", &mfma_throughput, py::arg("out"), py::arg("iters"),
        py::arg("shape") = 16);
  m.def("mfma_gemm16", &mfma_gemm16, py::arg("A"), py::arg("B"),
        py::arg("layout") = 0);
  m.def("mx_gemm16", &mx_gemm16, py::arg("A"), py::arg("B"),
        py::arg("fmt") = 0);
  m.def("has_mfma", &has_mfma);
}
