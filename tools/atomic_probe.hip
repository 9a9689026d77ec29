// atomic_probe — measure gfx950 groupby-accumulation strategies on the
// north-star shape (1e9 rows, 1e6-slot dense table) to pick the kernel
// design.  Standalone exe; results inform DESIGN.md.  Build:
//   hipcc --offload-arch=gfx950 -O3 tools/atomic_probe.hip -o tools/atomic_probe
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>
#include <vector>
#include <random>

#define CHECK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); exit(1); } } while (0)

constexpr int BLOCK = 256;

__device__ __forceinline__ int xcc_id() {
  // HW_REG_XCC_ID: s_getreg_b32 (hwRegId=29 on gfx9xx CDNA), 4 bits
  return __builtin_amdgcn_s_getreg(GETREG_IMMED(3, 0, 29)) & 0xF;
}

// v1: current production shape — f64 sum + u64 rowcnt, device-scope
__global__ void __launch_bounds__(BLOCK) k_v1(const int64_t* k, const double* v,
                                              int64_t n, int64_t slots,
                                              double* sums,
                                              unsigned long long* rowcnt) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int64_t s = k[i];
    atomicAdd(&rowcnt[s], 1ULL);
    double x = v[i];
    if (x == x) unsafeAtomicAdd(&sums[s], x);
  }
}

// v2: f64 sum only
__global__ void __launch_bounds__(BLOCK) k_v2(const int64_t* k, const double* v,
                                              int64_t n, int64_t slots,
                                              double* sums) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    double x = v[i];
    if (x == x) unsafeAtomicAdd(&sums[k[i]], x);
  }
}

// v3: u32 count only
__global__ void __launch_bounds__(BLOCK) k_v3(const int64_t* k, int64_t n,
                                              unsigned* rowcnt) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) atomicAdd(&rowcnt[k[i]], 1u);
}

// v4: per-XCD privatized tables, DEFAULT (device-scope) atomics
__global__ void __launch_bounds__(BLOCK) k_v4(const int64_t* k, const double* v,
                                              int64_t n, int64_t slots,
                                              double* sums8) {
  const int x = xcc_id();
  double* sums = sums8 + (int64_t)x * slots;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    double xv = v[i];
    if (xv == xv) unsafeAtomicAdd(&sums[k[i]], xv);
  }
}

// v5: per-XCD privatized tables, WORKGROUP-scope relaxed atomics (tests
// whether plain (non-sc) atomics run in the local XCD L2 at a higher rate)
__global__ void __launch_bounds__(BLOCK) k_v5(const int64_t* k, const double* v,
                                              int64_t n, int64_t slots,
                                              double* sums8) {
  const int x = xcc_id();
  double* sums = sums8 + (int64_t)x * slots;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    double xv = v[i];
    if (xv == xv)
      __hip_atomic_fetch_add(&sums[k[i]], xv, __ATOMIC_RELAXED,
                             __HIP_MEMORY_SCOPE_WORKGROUP);
  }
}

// v6: streaming ceiling — read keys+vals, no atomics
__global__ void __launch_bounds__(BLOCK) k_v6(const int64_t* k, const double* v,
                                              int64_t n, double* out) {
  double acc = 0;
  int64_t ks = 0;
  const int64_t npair = n >> 1;
  const longlong2* k2 = reinterpret_cast<const longlong2*>(k);
  const double2* v2 = reinterpret_cast<const double2*>(v);
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < npair; i += stride) {
    longlong2 kk = k2[i];
    double2 vv = v2[i];
    acc += vv.x + vv.y;
    ks += kk.x + kk.y;
  }
  if (acc == 1.2345 && ks == 42) out[threadIdx.x] = acc;  // never true
}

// v7: per-block LDS dense table (slots folded to 8192), ds f64 atomic adds +
// one global merge per block — the pass-2 aggregation shape
__global__ void __launch_bounds__(BLOCK) k_v7(const int64_t* k, const double* v,
                                              int64_t n, double* sums) {
  __shared__ double tab[8192];
  for (int s = threadIdx.x; s < 8192; s += blockDim.x) tab[s] = 0.0;
  __syncthreads();
  const int64_t npair = n >> 1;
  const longlong2* k2 = reinterpret_cast<const longlong2*>(k);
  const double2* v2 = reinterpret_cast<const double2*>(v);
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < npair; i += stride) {
    longlong2 kk = k2[i];
    double2 vv = v2[i];
    if (vv.x == vv.x) unsafeAtomicAdd(&tab[kk.x & 8191], vv.x);
    if (vv.y == vv.y) unsafeAtomicAdd(&tab[kk.y & 8191], vv.y);
  }
  __syncthreads();
  for (int s = threadIdx.x; s < 8192; s += blockDim.x)
    if (tab[s] != 0.0) unsafeAtomicAdd(&sums[s], tab[s]);
}

// v8: like v7 but per-WAVE u32 LDS histogram + rank (the scatter pass's LDS
// cost shape): hist count + compute rank, no global writes
__global__ void __launch_bounds__(BLOCK) k_v8(const int64_t* k, int64_t n,
                                              unsigned* out) {
  __shared__ unsigned hist[256];
  for (int s = threadIdx.x; s < 256; s += blockDim.x) hist[s] = 0;
  __syncthreads();
  unsigned acc = 0;
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    int b = (int)(k[i] >> 12) & 255;
    acc += atomicAdd(&hist[b], 1u);
  }
  if (acc == 0xFFFFFFFFu) out[threadIdx.x] = acc;
}

// v9: scattered 8B stores into 256 bucket streams (write-coalescing probe):
// position = running per-block cursor per bucket in LDS (not a correct
// multisplit, measures the memory pattern cost)
__global__ void __launch_bounds__(BLOCK) k_v9(const int64_t* k, const double* v,
                                              int64_t n, double* out,
                                              int64_t cap) {
  __shared__ unsigned cur[256];
  const unsigned base = (unsigned)((int64_t)blockIdx.x * (cap / gridDim.x));
  for (int s = threadIdx.x; s < 256; s += blockDim.x) cur[s] = 0;
  __syncthreads();
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int64_t percap = cap / gridDim.x / 256;
  for (; i < n; i += stride) {
    int b = (int)(k[i] >> 12) & 255;
    unsigned r = atomicAdd(&cur[b], 1u);
    out[base + (int64_t)b * percap + (r % percap)] = v[i];
  }
}

double bench(const char* name, int reps, void (*fn)(), double bytes) {
  // warm
  fn();
  CHECK(hipDeviceSynchronize());
  hipEvent_t a, b;
  CHECK(hipEventCreate(&a));
  CHECK(hipEventCreate(&b));
  CHECK(hipEventRecord(a));
  for (int r = 0; r < reps; ++r) fn();
  CHECK(hipEventRecord(b));
  CHECK(hipDeviceSynchronize());
  float ms = 0;
  CHECK(hipEventElapsedTime(&ms, a, b));
  double per = ms / reps;
  printf("%-34s %8.2f ms  %8.1f GB/s(alg16B) %8.1f Grows/s\n", name, per,
         bytes / per / 1e6, (bytes / 16.0) / per / 1e6);
  fflush(stdout);
  return per;
}

int main(int argc, char** argv) {
  int64_t n = argc > 1 ? atoll(argv[1]) : 1000000000LL;
  int64_t slots = 1000000;
  printf("n=%lld slots=%lld\n", (long long)n, (long long)slots);
  int64_t* k;
  double* v;
  double* sums;
  unsigned long long* rc;
  unsigned* rc32;
  double* sums8;
  double* scratch;
  CHECK(hipMalloc(&k, n * 8));
  CHECK(hipMalloc(&v, n * 8));
  CHECK(hipMalloc(&sums, slots * 8));
  CHECK(hipMalloc(&rc, slots * 8));
  CHECK(hipMalloc(&rc32, slots * 4));
  CHECK(hipMalloc(&sums8, 8 * slots * 8));
  CHECK(hipMalloc(&scratch, n * 8));
  // fill keys uniform via a tiny kernel-free path: host chunks
  {
    std::mt19937_64 gen(42);
    std::vector<int64_t> hk(1 << 24);
    std::vector<double> hv(1 << 24);
    for (int64_t off = 0; off < n; off += hk.size()) {
      int64_t m = std::min<int64_t>(hk.size(), n - off);
      for (int64_t i = 0; i < m; ++i) {
        hk[i] = gen() % slots;
        hv[i] = (double)(gen() % 1000) / 1000.0;
      }
      CHECK(hipMemcpy(k + off, hk.data(), m * 8, hipMemcpyHostToDevice));
      CHECK(hipMemcpy(v + off, hv.data(), m * 8, hipMemcpyHostToDevice));
    }
  }
  CHECK(hipMemset(sums, 0, slots * 8));
  CHECK(hipMemset(rc, 0, slots * 8));
  CHECK(hipMemset(rc32, 0, slots * 4));
  CHECK(hipMemset(sums8, 0, 8 * slots * 8));
  const int grid = 4096;
  double bytes = 16.0 * n;
  static int64_t N;
  static int64_t S;
  N = n;
  S = slots;
  static int64_t* K;
  static double* V;
  static double* SU;
  static unsigned long long* RC;
  static unsigned* RC32;
  static double* SU8;
  static double* SCR;
  K = k; V = v; SU = sums; RC = rc; RC32 = rc32; SU8 = sums8; SCR = scratch;

  bench("v6 streaming ceiling (16B/row)", 3, [] {
    hipLaunchKernelGGL(k_v6, dim3(4096), dim3(BLOCK), 0, 0, K, V, N, SCR);
  }, bytes);
  bench("v1 f64+u64 atomics (prod)", 2, [] {
    hipLaunchKernelGGL(k_v1, dim3(4096), dim3(BLOCK), 0, 0, K, V, N, S, SU, RC);
  }, bytes);
  bench("v2 f64 atomic only", 2, [] {
    hipLaunchKernelGGL(k_v2, dim3(4096), dim3(BLOCK), 0, 0, K, V, N, S, SU);
  }, bytes);
  bench("v3 u32 atomic only", 2, [] {
    hipLaunchKernelGGL(k_v3, dim3(4096), dim3(BLOCK), 0, 0, K, N, RC32);
  }, bytes);
  bench("v4 per-XCD tables, device scope", 2, [] {
    hipLaunchKernelGGL(k_v4, dim3(4096), dim3(BLOCK), 0, 0, K, V, N, S, SU8);
  }, bytes);
  bench("v5 per-XCD tables, wg scope", 2, [] {
    hipLaunchKernelGGL(k_v5, dim3(4096), dim3(BLOCK), 0, 0, K, V, N, S, SU8);
  }, bytes);
  bench("v7 LDS 8192-slot tables + merge", 3, [] {
    hipLaunchKernelGGL(k_v7, dim3(2048), dim3(BLOCK), 0, 0, K, V, N, SU);
  }, bytes);
  bench("v8 LDS hist256 rank (scatter A)", 3, [] {
    hipLaunchKernelGGL(k_v8, dim3(4096), dim3(BLOCK), 0, 0, K, N, RC32);
  }, bytes);
  bench("v9 256-stream scatter writes", 3, [] {
    hipLaunchKernelGGL(k_v9, dim3(1024), dim3(BLOCK), 0, 0, K, V, N, SCR, N);
  }, bytes);
  printf("done\n");
  return 0;
}
