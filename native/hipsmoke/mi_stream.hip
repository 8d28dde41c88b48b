// mi-stream — standalone CDNA4 STREAM + MFMA smoke payload for the
// validation pod.
//
// The MI355X-native replacement for the reference's in-pod `nvidia-smi`
// payload (/root/reference/nvidia-smi.yaml:13) with actual GPU work: an
// HBM3E STREAM suite (copy/scale/add/triad, plain + non-temporal) and an
// MFMA matrix-core warm-up that lights up all 8 XCDs, printing a
// golden-output-style table (the analog of README.md:137-156) plus one
// machine-readable JSON line. No torch dependency — links only the HIP
// runtime, so the smoke container stays small.
//
// Build: hipcc --offload-arch=gfx950 -O3 mi_stream.hip -o mi-stream

#include <chrono>
#include <cmath>
#include <cstdio>
#include <cstring>
#include <string>
#include <vector>

#include "../../k3samd/ops/hip/stream_kernels.h"
#include "../topology/gpu_health.h"
#include "../topology/kfd_topology.h"

#define HIP_CHECK(x)                                                         \
  do {                                                                       \
    hipError_t err_ = (x);                                                   \
    if (err_ != hipSuccess) {                                                \
      std::fprintf(stderr, "mi-stream: %s failed: %s\n", #x,                 \
                   hipGetErrorString(err_));                                 \
      return 1;                                                              \
    }                                                                        \
  } while (0)

using k3samd_kern::f4;

namespace {

struct Timing {
  float best_ms = 1e30f;
};

double gbps(double bytes, float ms) { return bytes / (ms * 1e6); }

// Numerics self-check: the pod log doubles as a correctness oracle (the
// reference's nvidia-smi output carried version oracles the same way,
// /root/reference/README.md:139-147). Three checks against host-computed
// references: fp32 triad (exact), one bf16 MFMA tile, one block-scaled
// MX-fp8 tile (identity-A => must reproduce B exactly).
bool self_check() {
  using namespace k3samd_kern;
  bool ok = true;
  // --- triad, 4096 elements, exactly-representable values
  {
    const int64_t n = 4096, n4 = n / 4;
    float *a, *b, *c;
    (void)hipMalloc(&a, n * 4);
    (void)hipMalloc(&b, n * 4);
    (void)hipMalloc(&c, n * 4);
    std::vector<float> hb(n), hc(n), ha(n);
    for (int64_t i = 0; i < n; ++i) {
      hb[i] = (float)((i % 31) - 15);
      hc[i] = (float)((i % 17) - 8) * 0.5f;
    }
    (void)hipMemcpy(b, hb.data(), n * 4, hipMemcpyHostToDevice);
    (void)hipMemcpy(c, hc.data(), n * 4, hipMemcpyHostToDevice);
    hipLaunchKernelGGL(stream_triad_kernel<true>, dim3(stream_grid(n4)),
                       dim3(kThreadsPerBlock), 0, 0, (f4*)a, (f4*)b, (f4*)c,
                       2.5f, n4);
    (void)hipMemcpy(ha.data(), a, n * 4, hipMemcpyDeviceToHost);
    for (int64_t i = 0; i < n && ok; ++i)
      if (ha[i] != hb[i] + 2.5f * hc[i]) ok = false;
    (void)hipFree(a); (void)hipFree(b); (void)hipFree(c);
    if (!ok) { std::printf("numerics: triad FAIL\n"); return false; }
  }
#if 1
  // --- bf16 MFMA tile vs host float reference
  double mfma_err = 0;
  {
    uint16_t *A, *B; float* D;
    (void)hipMalloc(&A, 16 * 32 * 2);
    (void)hipMalloc(&B, 32 * 16 * 2);
    (void)hipMalloc(&D, 16 * 16 * 4);
    std::vector<uint16_t> hA(16 * 32), hB(32 * 16);
    std::vector<float> fA(16 * 32), fB(32 * 16);
    auto to_bf16 = [](float f) -> uint16_t {
      uint32_t u;
      std::memcpy(&u, &f, 4);
      return (uint16_t)(u >> 16);  // values chosen exactly representable
    };
    for (int i = 0; i < 16 * 32; ++i) {
      fA[i] = (float)((i % 13) - 6) * 0.25f;
      hA[i] = to_bf16(fA[i]);
    }
    for (int i = 0; i < 32 * 16; ++i) {
      fB[i] = (float)((i % 11) - 5) * 0.5f;   // asymmetric vs A
      hB[i] = to_bf16(fB[i]);
    }
    (void)hipMemcpy(A, hA.data(), hA.size() * 2, hipMemcpyHostToDevice);
    (void)hipMemcpy(B, hB.data(), hB.size() * 2, hipMemcpyHostToDevice);
    hipLaunchKernelGGL(mfma_gemm16_kernel, dim3(1), dim3(64), 0, 0, A, B, D,
                       0);
    std::vector<float> hD(256);
    (void)hipMemcpy(hD.data(), D, 256 * 4, hipMemcpyDeviceToHost);
    for (int i = 0; i < 16; ++i)
      for (int j = 0; j < 16; ++j) {
        float ref = 0;
        for (int k = 0; k < 32; ++k) ref += fA[i * 32 + k] * fB[k * 16 + j];
        double e = std::abs(hD[i * 16 + j] - ref);
        if (e > mfma_err) mfma_err = e;
      }
    (void)hipFree(A); (void)hipFree(B); (void)hipFree(D);
    if (mfma_err > 1e-3) {
      std::printf("numerics: bf16 MFMA FAIL (err %g)\n", mfma_err);
      return false;
    }
  }
  // --- MX-fp8 identity tile: D must equal B rows exactly
  double mx_err = 0;
  {
    uint8_t *A, *B; float* D;
    (void)hipMalloc(&A, 16 * 128);
    (void)hipMalloc(&B, 16 * 128);
    (void)hipMalloc(&D, 16 * 16 * 4);
    std::vector<uint8_t> hA(16 * 128, 0), hB(16 * 128);
    for (int i = 0; i < 16; ++i) hA[i * 128 + i] = 0x38;  // e4m3 1.0 on diag
    // B values: exact small e4m3 codes, col-major pack, asymmetric
    std::vector<float> fB(128 * 16);
    const float lut[8] = {0.f, 0.5f, 1.f, 1.5f, 2.f, 3.f, 4.f, 6.f};
    for (int k = 0; k < 128; ++k)
      for (int j = 0; j < 16; ++j) {
        int code = (k * 7 + j * 3) % 16;  // uses sign bit too
        float v = lut[code & 7] * ((code & 8) ? -1.f : 1.f);
        fB[k * 16 + j] = v;
        // e4m3 encodings of the lut: 0,0x30,0x38,0x3c,0x40,0x44,0x48,0x4c
        const uint8_t enc[8] = {0, 0x30, 0x38, 0x3c, 0x40, 0x44, 0x48, 0x4c};
        hB[j * 128 + k] = enc[code & 7] | ((code & 8) ? 0x80 : 0);
      }
    (void)hipMemcpy(A, hA.data(), hA.size(), hipMemcpyHostToDevice);
    (void)hipMemcpy(B, hB.data(), hB.size(), hipMemcpyHostToDevice);
    hipLaunchKernelGGL(mx_gemm16_kernel<0>, dim3(1), dim3(64), 0, 0, A, B, D);
    std::vector<float> hD(256);
    (void)hipMemcpy(hD.data(), D, 256 * 4, hipMemcpyDeviceToHost);
    for (int i = 0; i < 16; ++i)
      for (int j = 0; j < 16; ++j) {
        double e = std::abs(hD[i * 16 + j] - fB[i * 16 + j]);
        if (e > mx_err) mx_err = e;
      }
    (void)hipFree(A); (void)hipFree(B); (void)hipFree(D);
    if (mx_err != 0.0) {
      std::printf("numerics: MX-fp8 FAIL (err %g)\n", mx_err);
      return false;
    }
  }
  std::printf(
      "numerics: triad exact, bf16 MFMA err %.2g, MX-fp8 identity exact\n",
      mfma_err);
#endif
  return true;
}

}  // namespace

int main(int argc, char** argv) {
  int64_t mib = 1024;
  int iters = 20;
  int device = 0;
  bool do_mfma = true;
  bool do_check = true;
  bool tune = false;
  bool lds_compare = false;
  bool all_gpus = false;
  int burn_s = 0;
  for (int i = 1; i < argc; ++i) {
    if (!std::strcmp(argv[i], "--mib") && i + 1 < argc)
      mib = std::atoll(argv[++i]);
    else if (!std::strcmp(argv[i], "--iters") && i + 1 < argc)
      iters = std::atoi(argv[++i]);
    else if (!std::strcmp(argv[i], "--device") && i + 1 < argc)
      device = std::atoi(argv[++i]);
    else if (!std::strcmp(argv[i], "--no-mfma"))
      do_mfma = false;
    else if (!std::strcmp(argv[i], "--no-check"))
      do_check = false;
    else if (!std::strcmp(argv[i], "--tune"))
      tune = true;
    else if (!std::strcmp(argv[i], "--lds-compare"))
      lds_compare = true;
    else if (!std::strcmp(argv[i], "--all-gpus"))
      all_gpus = true;
    else if (!std::strcmp(argv[i], "--burn") && i + 1 < argc)
      burn_s = std::atoi(argv[++i]);
    else {
      std::printf("mi-stream [--mib N] [--iters N] [--device D] [--no-mfma]"
                  " [--tune] [--lds-compare] [--all-gpus] [--burn SECONDS]\n");
      return !std::strcmp(argv[i], "--help") ? 0 : 2;
    }
  }

  int ndev = 0;
  HIP_CHECK(hipGetDeviceCount(&ndev));
  if (ndev == 0) {
    std::fprintf(stderr, "mi-stream: no GPUs visible\n");
    return 1;
  }

  if (all_gpus) {
    if (do_check && !self_check()) return 3;  // oracle on device 0 first
    // concurrent non-temporal triad on every visible GPU (the in-pod
    // demonstration of the headline metric at amd.com/gpu: N): per-GPU
    // buffers + stream, launches overlapped, one global wall-clock.
    const int64_t n_ = mib * (1 << 20) / 4;
    const int64_t n4_ = n_ / 4;
    const double step_bytes = 3.0 * n_ * 4;
    std::vector<f4*> A(ndev), B(ndev), C(ndev);
    std::vector<hipStream_t> streams(ndev);
    for (int d = 0; d < ndev; ++d) {
      HIP_CHECK(hipSetDevice(d));
      HIP_CHECK(hipMalloc(&A[d], n_ * 4));
      HIP_CHECK(hipMalloc(&B[d], n_ * 4));
      HIP_CHECK(hipMalloc(&C[d], n_ * 4));
      HIP_CHECK(hipMemset(B[d], 0x3c, n_ * 4));
      HIP_CHECK(hipMemset(C[d], 0x3d, n_ * 4));
      HIP_CHECK(hipStreamCreate(&streams[d]));
    }
    dim3 g((uint32_t)k3samd_kern::stream_grid(n4_));
    dim3 blk(k3samd_kern::kThreadsPerBlock);
    auto launch_all = [&] {
      for (int d = 0; d < ndev; ++d) {
        (void)hipSetDevice(d);
        k3samd_kern::launch_chunked(n4_, [&](int64_t off, int64_t cnt) {
          dim3 gc((uint32_t)k3samd_kern::stream_grid(cnt));
          hipLaunchKernelGGL(k3samd_kern::stream_triad_kernel<true>, gc, blk,
                             0, streams[d], A[d] + off, B[d] + off,
                             C[d] + off, 2.5f, cnt);
        });
      }
    };
    auto sync_all = [&] {
      for (int d = 0; d < ndev; ++d) (void)hipStreamSynchronize(streams[d]);
    };
    for (int w = 0; w < 3; ++w) launch_all();
    sync_all();
    auto t0 = std::chrono::steady_clock::now();
    for (int it = 0; it < iters; ++it) launch_all();
    sync_all();
    double sec = std::chrono::duration<double>(
                     std::chrono::steady_clock::now() - t0).count();
    double agg = ndev * step_bytes * iters / sec / 1e9;
    std::printf("mi-stream all-gpus: %d GPU(s) x %lld MiB, aggregate triad "
                "%.1f GB/s (%.1f per GPU)\n",
                ndev, (long long)mib, agg, agg / ndev);
    std::printf("{\"payload\": \"mi-stream\", \"mode\": \"all-gpus\", "
                "\"n_gpus\": %d, \"aggregate_triad_gbps\": %.1f}\n",
                ndev, agg);
    for (int d = 0; d < ndev; ++d) {
      (void)hipSetDevice(d);
      (void)hipFree(A[d]); (void)hipFree(B[d]); (void)hipFree(C[d]);
      (void)hipStreamDestroy(streams[d]);
    }
    return 0;
  }

  HIP_CHECK(hipSetDevice(device));
  hipDeviceProp_t prop;
  HIP_CHECK(hipGetDeviceProperties(&prop, device));

  const char* gpu_name =
      prop.name[0] ? prop.name : "AMD GPU";  // some boxes report no name
  std::printf("mi-stream: %s (%s), %d visible GPU(s), %d CUs, %.0f GiB VRAM\n",
              gpu_name, prop.gcnArchName, ndev, prop.multiProcessorCount,
              (double)prop.totalGlobalMem / (1 << 30));

  if (do_check && !self_check()) return 3;  // pod log is the oracle

  const int64_t n = mib * (1 << 20) / 4;  // fp32 elements
  const int64_t n4 = n / 4;
  const double buf_bytes = (double)n * 4;
  f4 *a, *b, *c;
  HIP_CHECK(hipMalloc(&a, buf_bytes));
  HIP_CHECK(hipMalloc(&b, buf_bytes));
  HIP_CHECK(hipMalloc(&c, buf_bytes));
  HIP_CHECK(hipMemset(b, 0x3c, buf_bytes));
  HIP_CHECK(hipMemset(c, 0x3d, buf_bytes));

  hipEvent_t ev0, ev1;
  HIP_CHECK(hipEventCreate(&ev0));
  HIP_CHECK(hipEventCreate(&ev1));
  dim3 grid((uint32_t)k3samd_kern::stream_grid(n4));
  dim3 block(k3samd_kern::kThreadsPerBlock);
  const float s = 2.5f;

  if (lds_compare) {
    // direct-nt vs LDS-staged triad (SURVEY §2c blueprint comparison)
    const double bytes = 3 * buf_bytes;
    auto time_k = [&](auto launch_fn) -> double {
      launch_fn();
      (void)hipDeviceSynchronize();
      float best = 1e30f;
      for (int it = 0; it < iters; ++it) {
        (void)hipEventRecord(ev0);
        launch_fn();
        (void)hipEventRecord(ev1);
        (void)hipEventSynchronize(ev1);
        float ms;
        (void)hipEventElapsedTime(&ms, ev0, ev1);
        if (ms < best) best = ms;
      }
      return gbps(bytes, best);
    };
    double direct = time_k([&] {
      hipLaunchKernelGGL(k3samd_kern::stream_triad_kernel<true>, grid, block,
                         0, 0, a, b, c, s, n4);
    });
    double lds = time_k([&] {
      hipLaunchKernelGGL(k3samd_kern::stream_triad_lds_kernel, grid, block,
                         0, 0, a, b, c, s, n4);
    });
    std::printf("triad %lld MiB: direct-nt %.1f GB/s | LDS-staged %.1f GB/s "
                "(%.1f%%)\n",
                (long long)mib, direct, lds, 100.0 * lds / direct);
    std::printf("{\"payload\": \"mi-stream-lds\", \"direct_gbps\": %.1f, "
                "\"lds_staged_gbps\": %.1f}\n", direct, lds);
    (void)hipFree(a); (void)hipFree(b); (void)hipFree(c);
    return 0;
  }

  if (tune) {
    // sweep block-size x unroll x grid-occupancy for the grid-stride
    // non-temporal triad; prints GB/s per config (exploration tool)
    const double bytes = 3 * buf_bytes;
    auto time_one = [&](auto kern, int tpb, int blocks) -> double {
      hipLaunchKernelGGL(kern, dim3(blocks), dim3(tpb), 0, 0, a, b, c, s, n4);
      hipLaunchKernelGGL(kern, dim3(blocks), dim3(tpb), 0, 0, a, b, c, s, n4);
      (void)hipDeviceSynchronize();
      float best = 1e30f;
      for (int it = 0; it < iters; ++it) {
        (void)hipEventRecord(ev0);
        hipLaunchKernelGGL(kern, dim3(blocks), dim3(tpb), 0, 0, a, b, c, s, n4);
        (void)hipEventRecord(ev1);
        (void)hipEventSynchronize(ev1);
        float ms;
        (void)hipEventElapsedTime(&ms, ev0, ev1);
        if (ms < best) best = ms;
      }
      return gbps(bytes, best);
    };
    std::printf("tune: grid-stride nt triad, %lld MiB buffers\n",
                (long long)mib);
    std::printf("%8s %6s %8s %10s\n", "tpb", "unroll", "blocks", "GB/s");
    for (int tpb : {256, 512, 1024}) {
      for (int wavesper : {4, 8, 16, 32}) {  // blocks = CUs*waves*64/tpb
        int blocks = 256 * wavesper * 64 / tpb;
        double g1 = time_one(k3samd_kern::stream_triad_gs_kernel<true, 1>,
                             tpb, blocks);
        double g2 = time_one(k3samd_kern::stream_triad_gs_kernel<true, 2>,
                             tpb, blocks);
        double g4 = time_one(k3samd_kern::stream_triad_gs_kernel<true, 4>,
                             tpb, blocks);
        std::printf("%8d %6d %8d  U1 %8.1f U2 %8.1f U4 %8.1f\n", tpb, 1,
                    blocks, g1, g2, g4);
      }
    }
    (void)hipFree(a); (void)hipFree(b); (void)hipFree(c);
    return 0;
  }

  if (burn_s > 0) {
    // Burn-in: saturate HBM (nt triad) and the matrix pipes (bf16 MFMA)
    // concurrently on two streams for --burn seconds, sampling thermals
    // from sysfs. Node-acceptance analog of gpu-burn.
    hipStream_t s_mem, s_mfma;
    HIP_CHECK(hipStreamCreate(&s_mem));
    HIP_CHECK(hipStreamCreate(&s_mfma));
    float* mf_out;
    const int mf_blocks = 1024;
    HIP_CHECK(hipMalloc(&mf_out, mf_blocks * sizeof(float)));
    // resolve our card index for sysfs sampling (best effort)
    int card = -1;
    {
      auto topo = k3samd::enumerate_topology(k3samd::default_sysfs_root());
      if (!topo.gpus.empty()) card = topo.gpus[0].card_index;
    }
    auto now = [] { return std::chrono::steady_clock::now(); };
    auto t_end = now() + std::chrono::seconds(burn_s);
    // RAS snapshot before the stress: ECC errors that APPEAR during the
    // burn are the node-acceptance failure signal
    k3samd::GpuHealthCounters ras0 =
        k3samd::read_gpu_health(k3samd::default_sysfs_root(), card);
    uint64_t triads = 0, mfmas = 0;
    long max_temp = -1, max_pw = -1;
    std::printf("burn: %d s of concurrent HBM streaming + bf16 MFMA\n",
                burn_s);
    std::printf("%6s %12s %12s %6s %7s %6s\n", "t(s)", "GB/s", "TFLOP/s",
                "temp", "power", "busy");
    auto t0 = now();
    auto next_report = t0 + std::chrono::seconds(2);
    uint64_t triads_last = 0, mfmas_last = 0;
    auto t_last = t0;
    while (now() < t_end) {
      for (int i = 0; i < 8; ++i) {
        hipLaunchKernelGGL(k3samd_kern::stream_triad_kernel<true>, grid,
                           block, 0, s_mem, a, b, c, s, n4);
        hipLaunchKernelGGL(k3samd_kern::mfma_throughput_kernel,
                           dim3(mf_blocks), block, 0, s_mfma, mf_out, 512);
        ++triads;
        ++mfmas;
      }
      HIP_CHECK(hipStreamSynchronize(s_mem));
      HIP_CHECK(hipStreamSynchronize(s_mfma));
      if (now() >= next_report) {
        auto st = k3samd::read_runtime_stats(k3samd::default_sysfs_root(),
                                             card);
        double dt = std::chrono::duration<double>(now() - t_last).count();
        double gbs = (triads - triads_last) * 3.0 * buf_bytes / dt / 1e9;
        double tf =
            (mfmas - mfmas_last) * mf_blocks * 4.0 * 4.0 * 16384.0 * 512.0 /
            dt / 1e12;
        if (st.temp_mc > max_temp) max_temp = st.temp_mc;
        if (st.power_uw > max_pw) max_pw = st.power_uw;
        std::printf("%6.0f %12.1f %12.1f %5ldC %6.0fW %5ld%%\n",
                    std::chrono::duration<double>(now() - t0).count(), gbs,
                    tf, st.temp_mc < 0 ? -1 : st.temp_mc / 1000,
                    st.power_uw < 0 ? -1.0 : st.power_uw / 1e6,
                    st.busy_percent);
        std::fflush(stdout);
        triads_last = triads;
        mfmas_last = mfmas;
        t_last = now();
        next_report += std::chrono::seconds(2);
      }
    }
    double total_s = std::chrono::duration<double>(now() - t0).count();
    double avg_gbs = triads * 3.0 * buf_bytes / total_s / 1e9;
    double avg_tf =
        mfmas * mf_blocks * 4.0 * 4.0 * 16384.0 * 512.0 / total_s / 1e12;
    k3samd::GpuHealthCounters ras1 =
        k3samd::read_gpu_health(k3samd::default_sysfs_root(), card);
    long d_ue = (ras0.ras_ue >= 0 && ras1.ras_ue >= 0)
                    ? ras1.ras_ue - ras0.ras_ue : -1;
    long d_ce = (ras0.ras_ce >= 0 && ras1.ras_ce >= 0)
                    ? ras1.ras_ce - ras0.ras_ce : -1;
    bool ras_clean = d_ue <= 0 && d_ce <= 0;
    if (ras0.ras_present)
      std::printf("burn RAS delta: ue %+ld ce %+ld (%s)\n", d_ue, d_ce,
                  ras_clean ? "clean" : "ERRORS DURING BURN");
    std::printf(
        "{\"payload\": \"mi-burn\", \"seconds\": %.0f, \"avg_triad_gbps\": "
        "%.1f, \"avg_mfma_tflops\": %.1f, \"max_temp_c\": %ld, "
        "\"max_power_w\": %.0f, \"ras_delta_ue\": %ld, "
        "\"ras_delta_ce\": %ld, \"ras_clean\": %s}\n",
        total_s, avg_gbs, avg_tf, max_temp < 0 ? -1 : max_temp / 1000,
        max_pw < 0 ? -1.0 : max_pw / 1e6, d_ue, d_ce,
        ras_clean ? "true" : "false");
    HIP_CHECK(hipFree(mf_out));
    HIP_CHECK(hipStreamDestroy(s_mem));
    HIP_CHECK(hipStreamDestroy(s_mfma));
    (void)hipFree(a); (void)hipFree(b); (void)hipFree(c);
    return 0;
  }

  struct Row {
    const char* name;
    double bytes;
    Timing plain, nt;
  };
  Row rows[] = {
      {"copy", 2 * buf_bytes, {}, {}},
      {"scale", 2 * buf_bytes, {}, {}},
      {"add", 3 * buf_bytes, {}, {}},
      {"triad", 3 * buf_bytes, {}, {}},
  };

  auto launch = [&](int op, bool nt) {
    using namespace k3samd_kern;
    // chunked: a flat launch caps at 2^32-1 work-items (AQL grid_size_x),
    // so buffers >= 64 GiB need more than one dispatch
    launch_chunked(n4, [&](int64_t off, int64_t cnt) {
      dim3 g((uint32_t)stream_grid(cnt));
      if (nt) {
        switch (op) {
          case 0: hipLaunchKernelGGL(stream_copy_kernel<true>, g, block, 0, 0, a + off, b + off, cnt); break;
          case 1: hipLaunchKernelGGL(stream_scale_kernel<true>, g, block, 0, 0, a + off, c + off, s, cnt); break;
          case 2: hipLaunchKernelGGL(stream_add_kernel<true>, g, block, 0, 0, a + off, b + off, c + off, cnt); break;
          case 3: hipLaunchKernelGGL(stream_triad_kernel<true>, g, block, 0, 0, a + off, b + off, c + off, s, cnt); break;
        }
      } else {
        switch (op) {
          case 0: hipLaunchKernelGGL(stream_copy_kernel<false>, g, block, 0, 0, a + off, b + off, cnt); break;
          case 1: hipLaunchKernelGGL(stream_scale_kernel<false>, g, block, 0, 0, a + off, c + off, s, cnt); break;
          case 2: hipLaunchKernelGGL(stream_add_kernel<false>, g, block, 0, 0, a + off, b + off, c + off, cnt); break;
          case 3: hipLaunchKernelGGL(stream_triad_kernel<false>, g, block, 0, 0, a + off, b + off, c + off, s, cnt); break;
        }
      }
    });
  };

  for (int op = 0; op < 4; ++op) {
    for (int nt = 0; nt < 2; ++nt) {
      launch(op, nt);  // warm
      launch(op, nt);
      HIP_CHECK(hipDeviceSynchronize());
      Timing& t = nt ? rows[op].nt : rows[op].plain;
      for (int it = 0; it < iters; ++it) {
        HIP_CHECK(hipEventRecord(ev0));
        launch(op, nt);
        HIP_CHECK(hipEventRecord(ev1));
        HIP_CHECK(hipEventSynchronize(ev1));
        float ms;
        HIP_CHECK(hipEventElapsedTime(&ms, ev0, ev1));
        if (ms < t.best_ms) t.best_ms = ms;
      }
    }
  }

  std::printf("+--------+-------------------+-------------------+\n");
  std::printf("| STREAM | plain GB/s        | non-temporal GB/s |\n");
  std::printf("+--------+-------------------+-------------------+\n");
  for (auto& r : rows) {
    std::printf("| %-6s | %17.1f | %17.1f |\n", r.name,
                gbps(r.bytes, r.plain.best_ms), gbps(r.bytes, r.nt.best_ms));
  }
  std::printf("+--------+-------------------+-------------------+\n");

  double mfma_tf = 0, mfma32_tf = 0;
  if (do_mfma) {
    // 4096 blocks + 4 chains measured best (tools/mfma_tune.hip sweep:
    // 2477 TF = 99 % of the 2.5 PF dense peak)
    const int blocks = 4096, mf_iters = 4096;
    float* out;
    HIP_CHECK(hipMalloc(&out, blocks * sizeof(float)));
    auto time_mfma = [&](auto kern, double flops_per_wave_iter,
                         int accs) -> double {
      hipLaunchKernelGGL(kern, dim3(blocks), block, 0, 0, out, 256);  // warm
      (void)hipDeviceSynchronize();
      float best = 1e30f;
      for (int rep = 0; rep < 3; ++rep) {  // best-of-3: ride out clock ramp
        (void)hipEventRecord(ev0);
        hipLaunchKernelGGL(kern, dim3(blocks), block, 0, 0, out, mf_iters);
        (void)hipEventRecord(ev1);
        (void)hipEventSynchronize(ev1);
        float ms;
        (void)hipEventElapsedTime(&ms, ev0, ev1);
        if (ms < best) best = ms;
      }
      double flops =
          (double)blocks * 4 * accs * flops_per_wave_iter * mf_iters;
      return flops / (best * 1e9);
    };
    mfma_tf = time_mfma(k3samd_kern::mfma_throughput_kernel, 16384.0, 4);
    mfma32_tf = time_mfma(k3samd_kern::mfma_throughput32_kernel, 32768.0, 4);
    double mx8_tf =
        time_mfma(k3samd_kern::mfma_throughput_mx_kernel<0>, 65536.0, 4);
    double mx6_tf =
        time_mfma(k3samd_kern::mfma_throughput_mx_kernel<2>, 65536.0, 4);
    double mx4_tf =
        time_mfma(k3samd_kern::mfma_throughput_mx_kernel<4>, 65536.0, 4);
    std::printf("| MFMA bf16 16x16x32: %7.1f TF | 32x32x16: %7.1f TF     |\n",
                mfma_tf, mfma32_tf);
    std::printf("| MFMA MX 16x16x128  fp8: %6.1f | fp6: %6.1f | fp4: %6.1f |\n",
                mx8_tf, mx6_tf, mx4_tf);
    std::printf("+--------+-------------------+-------------------+\n");
    if (mfma32_tf > mfma_tf) mfma_tf = mfma32_tf;
    HIP_CHECK(hipFree(out));
  }

  double triad_best =
      gbps(rows[3].bytes, std::min(rows[3].plain.best_ms, rows[3].nt.best_ms));
  std::printf(
      "{\"payload\": \"mi-stream\", \"gpu\": \"%s\", \"arch\": \"%s\", "
      "\"buffer_MiB\": %lld, \"triad_gbps\": %.1f, \"mfma_bf16_tflops\": "
      "%.1f}\n",
      gpu_name, prop.gcnArchName, (long long)mib, triad_best, mfma_tf);

  HIP_CHECK(hipFree(a));
  HIP_CHECK(hipFree(b));
  HIP_CHECK(hipFree(c));
  return 0;
}
