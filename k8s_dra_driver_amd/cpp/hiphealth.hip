// _hiphealth — gfx950 GPU health-check & demo workload kernels.
//
// The reference ships NVIDIA's prebuilt nbody sample as its demo/sharing
// workload (demo/specs/quickstart/gpu-test5.yaml:53-87); this module is the
// MI355X-native equivalent: a small set of CDNA4 kernels
// the driver's GPU tests, smoke path and sharing demos run on prepared
// devices:
//
//   bandwidth_gbs()  - float4 streaming copy (HBM3E bandwidth probe)
//   mfma_check()     - v_mfma_f32_16x16x32_bf16 correctness (matrix pipes)
//   mfma_tflops()    - bf16 MFMA throughput burn (8 independent
//                      accumulators to cover the ~17 cyc/SIMD issue rate)
//   burn_ms()        - VALU occupancy burner for time-slice/share demos
//
// Written for gfx950: 64-wide waves, 256 CUs in 8 XCDs (grids sized
// >> 256 workgroups), 16 B/lane streaming accesses.
// Build: hipcc --offload-arch=gfx950 (driven by setup.py / build()).

#include <hip/hip_runtime.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <cmath>
#include <stdexcept>
#include <string>
#include <vector>

namespace py = pybind11;

#define HIP_CHECK(expr)                                                    \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess)                                                  \
      throw std::runtime_error(std::string(#expr) + ": " +                 \
                               hipGetErrorString(_e));                     \
  } while (0)

namespace {

// ---------------------------------------------------------------------------
// bandwidth: float4 streaming copy
// ---------------------------------------------------------------------------
__global__ void copy_f4(const float4* __restrict__ src,
                        float4* __restrict__ dst, size_t n) {
  size_t i = blockIdx.x * (size_t)blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) dst[i] = src[i];
}

// Non-temporal variant: streamed data is read/written once — bypassing
// cache retention buys back bandwidth on pure streams (the guide's
// nt-weights pattern applied to a copy).
typedef float vf4 __attribute__((ext_vector_type(4)));  // nt builtins need
                                                        // a native vector

__global__ void copy_f4_nt(const float4* __restrict__ src,
                           float4* __restrict__ dst, size_t n) {
  const vf4* __restrict__ s = reinterpret_cast<const vf4*>(src);
  vf4* __restrict__ d = reinterpret_cast<vf4*>(dst);
  size_t i = blockIdx.x * (size_t)blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    vf4 v = __builtin_nontemporal_load(&s[i]);
    __builtin_nontemporal_store(v, &d[i]);
  }
}

// 4-way grid-stride unroll: four independent coalesced accesses in
// flight per lane (guide: keep >=8 loads/lane outstanding on streams).
__global__ void copy_f4_nt_u4(const float4* __restrict__ src,
                              float4* __restrict__ dst, size_t n) {
  const vf4* __restrict__ s = reinterpret_cast<const vf4*>(src);
  vf4* __restrict__ d = reinterpret_cast<vf4*>(dst);
  size_t i = blockIdx.x * (size_t)blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i + 3 * stride < n; i += 4 * stride) {
    vf4 a = __builtin_nontemporal_load(&s[i]);
    vf4 b = __builtin_nontemporal_load(&s[i + stride]);
    vf4 c = __builtin_nontemporal_load(&s[i + 2 * stride]);
    vf4 e = __builtin_nontemporal_load(&s[i + 3 * stride]);
    __builtin_nontemporal_store(a, &d[i]);
    __builtin_nontemporal_store(b, &d[i + stride]);
    __builtin_nontemporal_store(c, &d[i + 2 * stride]);
    __builtin_nontemporal_store(e, &d[i + 3 * stride]);
  }
  for (; i < n; i += stride) {
    vf4 v = __builtin_nontemporal_load(&s[i]);
    __builtin_nontemporal_store(v, &d[i]);
  }
}

static double time_copy(void (*kernel)(const float4*, float4*, size_t),
                        const float4* src, float4* dst, size_t n,
                        unsigned blocks, int iters, size_t bytes) {
  dim3 block(256), grid(blocks);
  hipLaunchKernelGGL(kernel, grid, block, 0, 0, src, dst, n);  // warmup
  HIP_CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  HIP_CHECK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i)
    hipLaunchKernelGGL(kernel, grid, block, 0, 0, src, dst, n);
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  double gb = 2.0 * (double)bytes * iters / 1e9;  // read + write
  return gb / (ms / 1e3);
}

double bandwidth_gbs(int device, int mib, int iters, unsigned blocks,
                     bool nt) {
  HIP_CHECK(hipSetDevice(device));
  size_t bytes = (size_t)mib << 20;
  size_t n = bytes / sizeof(float4);
  float4 *src = nullptr, *dst = nullptr;
  HIP_CHECK(hipMalloc(&src, bytes));
  HIP_CHECK(hipMalloc(&dst, bytes));
  HIP_CHECK(hipMemset(src, 1, bytes));
  // defaults from the round-1 hardware sweeps (profiles/): 1024 WGs
  // (4 per CU) beats every larger grid; plain and nt tie within run
  // variance at that size (5495-5590 GB/s) and the 4-way unroll LOSES
  // ~10% there (measured 4965), so the default is the simple nt loop.
  if (blocks == 0) blocks = 1024;
  double gbs = time_copy(nt ? copy_f4_nt : copy_f4, src, dst, n, blocks,
                         iters, bytes);
  HIP_CHECK(hipFree(src));
  HIP_CHECK(hipFree(dst));
  return gbs;
}

py::dict bandwidth_sweep(int device, int mib, int iters) {
  // variant sweep used to pick the probe's defaults on real hardware
  HIP_CHECK(hipSetDevice(device));
  size_t bytes = (size_t)mib << 20;
  size_t n = bytes / sizeof(float4);
  float4 *src = nullptr, *dst = nullptr;
  HIP_CHECK(hipMalloc(&src, bytes));
  HIP_CHECK(hipMalloc(&dst, bytes));
  HIP_CHECK(hipMemset(src, 1, bytes));
  py::dict out;
  for (unsigned blocks : {512u, 1024u, 2048u, 4096u}) {
    out[py::str("plain_" + std::to_string(blocks))] =
        time_copy(copy_f4, src, dst, n, blocks, iters, bytes);
    out[py::str("nt_" + std::to_string(blocks))] =
        time_copy(copy_f4_nt, src, dst, n, blocks, iters, bytes);
    out[py::str("ntu4_" + std::to_string(blocks))] =
        time_copy(copy_f4_nt_u4, src, dst, n, blocks, iters, bytes);
  }
  HIP_CHECK(hipFree(src));
  HIP_CHECK(hipFree(dst));
  return out;
}

// ---------------------------------------------------------------------------
// MFMA: v_mfma_f32_16x16x32_bf16
// ---------------------------------------------------------------------------
typedef short bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr short kBf16One = 0x3F80;  // 1.0 in bf16 bit pattern

__global__ void mfma_ones(float* out, int iters) {
  bf16x8 a, b;
#pragma unroll
  for (int k = 0; k < 8; ++k) {
    a[k] = kBf16One;
    b[k] = kBf16One;
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int i = 0; i < iters; ++i)
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  int lane = threadIdx.x & 63;
  size_t base = ((size_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) out[base + r] = acc[r];
}

// Throughput burn: 8 independent accumulators so back-to-back issue is not
// dependency-limited (~17 cyc/SIMD issue for 16x16x32).
__global__ void __launch_bounds__(256, 4) mfma_burn(float* sink, int iters) {
  bf16x8 a, b;
#pragma unroll
  for (int k = 0; k < 8; ++k) {
    a[k] = kBf16One;
    b[k] = (short)(kBf16One + (threadIdx.x & 1));
  }
  f32x4 acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = {0.f, 0.f, 0.f, 0.f};
  for (int i = 0; i < iters; ++i) {
#pragma unroll
    for (int j = 0; j < 8; ++j)
      acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[j], 0, 0, 0);
  }
  float s = 0;
#pragma unroll
  for (int j = 0; j < 8; ++j) s += acc[j][0] + acc[j][1] + acc[j][2] + acc[j][3];
  if (s == -1.f) sink[blockIdx.x] = s;  // never true; defeats DCE
}

py::dict mfma_check(int device) {
  HIP_CHECK(hipSetDevice(device));
  const int iters = 4;
  const int block = 256, grid = 64;
  size_t n = (size_t)block * grid * 4;
  float* out = nullptr;
  HIP_CHECK(hipMalloc(&out, n * sizeof(float)));
  hipLaunchKernelGGL(mfma_ones, dim3(grid), dim3(block), 0, 0, out, iters);
  HIP_CHECK(hipDeviceSynchronize());
  std::vector<float> host(n);
  HIP_CHECK(hipMemcpy(host.data(), out, n * sizeof(float),
                      hipMemcpyDeviceToHost));
  HIP_CHECK(hipFree(out));
  // ones(16x32) x ones(32x16): every element = K = 32 per iteration.
  const float want = 32.0f * iters;
  size_t bad = 0;
  for (float v : host)
    if (v != want) ++bad;
  py::dict d;
  d["ok"] = (bad == 0);
  d["expected"] = want;
  d["mismatches"] = bad;
  d["checked"] = n;
  return d;
}

double mfma_tflops(int device, int iters, int blocks) {
  HIP_CHECK(hipSetDevice(device));
  float* sink = nullptr;
  HIP_CHECK(hipMalloc(&sink, blocks * sizeof(float)));
  dim3 grid(blocks), block(256);
  hipLaunchKernelGGL(mfma_burn, grid, block, 0, 0, sink, 64);  // warmup
  HIP_CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  HIP_CHECK(hipEventRecord(t0));
  hipLaunchKernelGGL(mfma_burn, grid, block, 0, 0, sink, iters);
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  HIP_CHECK(hipFree(sink));
  // flops = 2*M*N*K per MFMA; 4 waves/block, 8 MFMAs per inner iter.
  double mfmas = (double)blocks * (256 / 64) * 8 * iters;
  double flops = mfmas * 2.0 * 16 * 16 * 32;
  return flops / (ms / 1e3) / 1e12;
}

// ---------------------------------------------------------------------------
// n-body (the reference's demo workload, gpu-test5.yaml:57 runs NVIDIA's
// nbody sample with --benchmark; this is the gfx950-native equivalent):
// all-pairs gravitation, bodies tiled through LDS so each position is
// fetched from HBM once per tile instead of once per pair.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(256) nbody_step(
    const float4* __restrict__ pos_in, float4* __restrict__ pos_out,
    float4* __restrict__ vel, int n, float dt, float softening2) {
  __shared__ float4 tile[256];
  int gid = blockIdx.x * blockDim.x + threadIdx.x;
  float4 my = pos_in[gid < n ? gid : 0];
  float ax = 0.f, ay = 0.f, az = 0.f;
  for (int base = 0; base < n; base += 256) {
    int j = base + threadIdx.x;
    tile[threadIdx.x] = pos_in[j < n ? j : 0];
    __syncthreads();
    int limit = min(256, n - base);
#pragma unroll 8
    for (int k = 0; k < limit; ++k) {
      float4 other = tile[k];
      float dx = other.x - my.x;
      float dy = other.y - my.y;
      float dz = other.z - my.z;
      float r2 = dx * dx + dy * dy + dz * dz + softening2;
      float inv_r = __frsqrt_rn(r2);
      float f = other.w * inv_r * inv_r * inv_r;  // m / r^3
      ax = fmaf(f, dx, ax);
      ay = fmaf(f, dy, ay);
      az = fmaf(f, dz, az);
    }
    __syncthreads();
  }
  if (gid < n) {
    float4 v = vel[gid];
    v.x = fmaf(ax, dt, v.x);
    v.y = fmaf(ay, dt, v.y);
    v.z = fmaf(az, dt, v.z);
    vel[gid] = v;
    my.x = fmaf(v.x, dt, my.x);
    my.y = fmaf(v.y, dt, my.y);
    my.z = fmaf(v.z, dt, my.z);
    pos_out[gid] = my;
  }
}

// Positions after `iters` steps (first `sample` bodies) so tests can
// cross-check the integration against a CPU fp32 reference of the same
// deterministic init — a correctness artifact, not just finiteness
// (round-1 weak finding on nbody_benchmark).
py::list nbody_positions(int device, int num_bodies, int iters, int sample) {
  HIP_CHECK(hipSetDevice(device));
  int n = ((num_bodies + 255) / 256) * 256;
  size_t bytes = (size_t)n * sizeof(float4);
  float4 *pos_a = nullptr, *pos_b = nullptr, *vel = nullptr;
  HIP_CHECK(hipMalloc(&pos_a, bytes));
  HIP_CHECK(hipMalloc(&pos_b, bytes));
  HIP_CHECK(hipMalloc(&vel, bytes));
  std::vector<float4> host(n);
  unsigned s = 0x5a1ad;
  for (int i = 0; i < n; ++i) {
    auto rnd = [&s]() {
      s = s * 1664525u + 1013904223u;
      return (float)(s >> 8) / (float)(1u << 24) - 0.5f;
    };
    host[i] = make_float4(rnd(), rnd(), rnd(), 1.0f / n);
  }
  HIP_CHECK(hipMemcpy(pos_a, host.data(), bytes, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemset(vel, 0, bytes));
  dim3 block(256), grid(n / 256);
  const float dt = 1e-3f, soft2 = 1e-4f;
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(nbody_step, grid, block, 0, 0, pos_a, pos_b, vel, n,
                       dt, soft2);
    std::swap(pos_a, pos_b);
  }
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipMemcpy(host.data(), pos_a, bytes, hipMemcpyDeviceToHost));
  HIP_CHECK(hipFree(pos_a));
  HIP_CHECK(hipFree(pos_b));
  HIP_CHECK(hipFree(vel));
  py::list out;
  int m = sample < n ? sample : n;
  for (int i = 0; i < m; ++i) {
    py::tuple t = py::make_tuple(host[i].x, host[i].y, host[i].z);
    out.append(t);
  }
  return out;
}

py::dict nbody_benchmark(int device, int num_bodies, int iters) {
  HIP_CHECK(hipSetDevice(device));
  int n = ((num_bodies + 255) / 256) * 256;
  size_t bytes = (size_t)n * sizeof(float4);
  float4 *pos_a = nullptr, *pos_b = nullptr, *vel = nullptr;
  HIP_CHECK(hipMalloc(&pos_a, bytes));
  HIP_CHECK(hipMalloc(&pos_b, bytes));
  HIP_CHECK(hipMalloc(&vel, bytes));
  // deterministic pseudo-random init on host
  std::vector<float4> host(n);
  unsigned s = 0x5a1ad;
  for (int i = 0; i < n; ++i) {
    auto rnd = [&s]() {
      s = s * 1664525u + 1013904223u;
      return (float)(s >> 8) / (float)(1u << 24) - 0.5f;
    };
    host[i] = make_float4(rnd(), rnd(), rnd(), 1.0f / n);
  }
  HIP_CHECK(hipMemcpy(pos_a, host.data(), bytes, hipMemcpyHostToDevice));
  HIP_CHECK(hipMemset(vel, 0, bytes));
  dim3 block(256), grid(n / 256);
  const float dt = 1e-3f, soft2 = 1e-4f;
  hipLaunchKernelGGL(nbody_step, grid, block, 0, 0, pos_a, pos_b, vel, n,
                     dt, soft2);  // warmup
  HIP_CHECK(hipDeviceSynchronize());
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  HIP_CHECK(hipEventRecord(t0));
  for (int i = 0; i < iters; ++i) {
    hipLaunchKernelGGL(nbody_step, grid, block, 0, 0, pos_a, pos_b, vel, n,
                       dt, soft2);
    std::swap(pos_a, pos_b);
  }
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  // sanity: positions remain finite
  HIP_CHECK(hipMemcpy(host.data(), pos_a, bytes, hipMemcpyDeviceToHost));
  bool finite = true;
  for (int i = 0; i < n; i += 97)
    if (!std::isfinite(host[i].x)) finite = false;
  HIP_CHECK(hipFree(pos_a));
  HIP_CHECK(hipFree(pos_b));
  HIP_CHECK(hipFree(vel));
  // ~20 flops per pair interaction (nbody convention), n^2 pairs per step
  double gflops =
      20.0 * (double)n * n * iters / ((double)ms * 1e-3) / 1e9;
  py::dict out;
  out["bodies"] = n;
  out["iters"] = iters;
  out["ms_total"] = ms;
  out["gflops"] = gflops;
  out["finite"] = finite;
  return out;
}

// ---------------------------------------------------------------------------
// burner for sharing demos
// ---------------------------------------------------------------------------
__global__ void valu_burn(float* sink, long long iters) {
  float x = 1.0f + threadIdx.x * 1e-6f;
  for (long long i = 0; i < iters; ++i) x = fmaf(x, 1.0000001f, 1e-7f);
  if (x == -1.f) sink[blockIdx.x] = x;
}

double burn_ms(int device, int millis) {
  HIP_CHECK(hipSetDevice(device));
  float* sink = nullptr;
  HIP_CHECK(hipMalloc(&sink, 4096 * sizeof(float)));
  // calibrate: ~2 cycles per fma at 2.4 GHz -> ~1.2e6 iters/ms
  long long iters = (long long)millis * 1200000LL;
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  HIP_CHECK(hipEventRecord(t0));
  hipLaunchKernelGGL(valu_burn, dim3(1024), dim3(256), 0, 0, sink, iters);
  HIP_CHECK(hipEventRecord(t1));
  HIP_CHECK(hipEventSynchronize(t1));
  float ms = 0;
  HIP_CHECK(hipEventElapsedTime(&ms, t0, t1));
  HIP_CHECK(hipEventDestroy(t0));
  HIP_CHECK(hipEventDestroy(t1));
  HIP_CHECK(hipFree(sink));
  return (double)ms;
}

// ---------------------------------------------------------------------------
// device info
// ---------------------------------------------------------------------------
int device_count() {
  int n = 0;
  hipError_t e = hipGetDeviceCount(&n);
  if (e != hipSuccess) return 0;
  return n;
}

py::dict device_info(int device) {
  hipDeviceProp_t prop;
  HIP_CHECK(hipGetDeviceProperties(&prop, device));
  py::dict d;
  d["name"] = std::string(prop.name);
  d["gcn_arch"] = std::string(prop.gcnArchName);
  d["total_mem_mib"] = (size_t)(prop.totalGlobalMem >> 20);
  d["multi_processor_count"] = prop.multiProcessorCount;
  d["warp_size"] = prop.warpSize;
  d["max_threads_per_cu"] = prop.maxThreadsPerMultiProcessor;
  return d;
}

}  // namespace

PYBIND11_MODULE(_hiphealth, m) {
  m.doc() = "gfx950 health-check and demo workload kernels";
  m.def("device_count", &device_count);
  m.def("device_info", &device_info, py::arg("device") = 0);
  m.def("bandwidth_gbs", &bandwidth_gbs, py::arg("device") = 0,
        py::arg("mib") = 1024, py::arg("iters") = 10,
        py::arg("blocks") = 0, py::arg("nt") = true);
  m.def("bandwidth_sweep", &bandwidth_sweep, py::arg("device") = 0,
        py::arg("mib") = 1024, py::arg("iters") = 5);
  m.def("mfma_check", &mfma_check, py::arg("device") = 0);
  m.def("mfma_tflops", &mfma_tflops, py::arg("device") = 0,
        py::arg("iters") = 8192, py::arg("blocks") = 2048);
  m.def("burn_ms", &burn_ms, py::arg("device") = 0, py::arg("millis") = 100);
  m.def("nbody_positions", &nbody_positions, py::arg("device") = 0,
        py::arg("num_bodies") = 4096, py::arg("iters") = 3,
        py::arg("sample") = 64,
        "Body positions after iters steps (CPU cross-check hook)");
  m.def("nbody_benchmark", &nbody_benchmark, py::arg("device") = 0,
        py::arg("num_bodies") = 65536, py::arg("iters") = 10);
}
