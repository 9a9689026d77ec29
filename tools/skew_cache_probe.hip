// skew_cache_probe — round-2 roadmap item 1b: does an LDS
// aggregation cache in front of the groupby accumulate pay on the
// BASELINE zipf(1.2) key distribution while costing ~nothing on uniform?
//
// Hypothesis: on zipf(1.2) a ~4K-entry per-block (key, sum, rowcnt)
// direct-mapped cache absorbs most rows (head keys dominate), collapsing
// the global-atomic rate (the uniform-keys bottleneck, 23-27 G op/s
// chip-wide — tools/atomic_probe.hip) by the hit rate; on uniform 1e6
// keys the hit rate is ~cache/keys and the cache is pure overhead, so the
// variant must be gated by the cardinality/skew estimate the router
// already computes.
//
// Variants (all produce the dense (sums, rowcnt) tables; checked against
// a host reference):
//   v1  global-atomic accumulate (production k_gb_accum shape; baseline)
//   v2  LDS direct-mapped cache (key & (C-1)); occupied-slot conflicts
//       BYPASS to the global table (no eviction — tags are write-once, so
//       the kernel stays race-free without slot locks); flush at the end.
// If v2 pays, follow-ups to measure live: 2-way associativity with an
// LFU-ish evict, and wiring the cache into gb_scatter's front end.
// Run on 1 GPU:  ./skew_cache_probe [rows] [keys]
// Build: hipcc --offload-arch=gfx950 -O3 tools/skew_cache_probe.hip \
//        -o tools/skew_cache_probe
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>
#include <cmath>
#include <random>
#include <vector>

#define CHECK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); exit(1); } } while (0)

constexpr int BLOCK = 256;
constexpr int CACHE = 4096;           // LDS entries: 4096*(8+8+4) = 80 KB

__global__ void __launch_bounds__(BLOCK) k_v1(
    const int64_t* __restrict__ k, const double* __restrict__ v, int64_t n,
    double* __restrict__ sums, unsigned long long* __restrict__ rowcnt) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int64_t s = k[i];
    atomicAdd(&rowcnt[s], 1ULL);
    unsafeAtomicAdd(&sums[s], v[i]);
  }
}

// Direct-mapped LDS cache.  Entries are claimed per-slot with an atomicCAS
// on the tag; aggregation into a claimed slot is LDS-atomic (ds_add).
// Conflicting keys evict by flushing the resident entry's (sum, cnt) to
// the global table and re-claiming.  Tag -1 = empty.
__global__ void __launch_bounds__(BLOCK) k_v2(
    const int64_t* __restrict__ k, const double* __restrict__ v, int64_t n,
    double* __restrict__ sums, unsigned long long* __restrict__ rowcnt) {
  __shared__ long long tag[CACHE];
  __shared__ double lsum[CACHE];
  __shared__ unsigned lcnt[CACHE];
  for (int t = threadIdx.x; t < CACHE; t += blockDim.x) {
    tag[t] = -1;
    lsum[t] = 0.0;
    lcnt[t] = 0;
  }
  __syncthreads();
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const long long key = k[i];
    const double x = v[i];
    const int slot = (int)(key & (CACHE - 1));
    const long long cur = tag[slot];
    if (cur == key) {              // hit: LDS aggregate
      atomicAdd(&lsum[slot], x);
      atomicAdd(&lcnt[slot], 1u);
      continue;
    }
    if (cur == -1 &&
        atomicCAS((unsigned long long*)&tag[slot],
                  (unsigned long long)-1LL,
                  (unsigned long long)key) == (unsigned long long)-1LL) {
      atomicAdd(&lsum[slot], x);   // claimed empty slot
      atomicAdd(&lcnt[slot], 1u);
      continue;
    }
    // miss on an occupied slot: bypass straight to the global table (an
    // in-place evict+swap needs a slot lock — measure the simple form
    // first; the hit rate is what the probe is after)
    atomicAdd(&rowcnt[key], 1ULL);
    unsafeAtomicAdd(&sums[key], x);
  }
  __syncthreads();
  for (int t = threadIdx.x; t < CACHE; t += blockDim.x) {
    const long long key = tag[t];
    if (key >= 0) {
      if (lcnt[t]) atomicAdd(&rowcnt[key], (unsigned long long)lcnt[t]);
      if (lsum[t] != 0.0 || lcnt[t]) unsafeAtomicAdd(&sums[key], lsum[t]);
    }
  }
}

static void fill_zipf(std::vector<int64_t>& k, int64_t keys, double s,
                      std::mt19937_64& gen) {
  // bounded zipf via inverse-CDF over precomputed harmonic weights of the
  // first `keys` ranks; rank -> key id is an identity permutation (rank
  // locality in the key space is what the direct-mapped cache sees in the
  // BASELINE generator too: zipf ranks are small ints)
  std::vector<double> cdf(keys);
  double acc = 0.0;
  for (int64_t r = 0; r < keys; ++r) {
    acc += 1.0 / std::pow((double)(r + 1), s);
    cdf[r] = acc;
  }
  std::uniform_real_distribution<double> U(0.0, acc);
  for (auto& x : k) {
    const double u = U(gen);
    x = (int64_t)(std::lower_bound(cdf.begin(), cdf.end(), u) - cdf.begin());
  }
}

int main(int argc, char** argv) {
  const int64_t n = argc > 1 ? atoll(argv[1]) : 500'000'000LL;
  const int64_t keys = argc > 2 ? atoll(argv[2]) : 1'000'000LL;
  std::mt19937_64 gen(42);
  std::vector<int64_t> hk(n);
  std::vector<double> hv(n);
  std::uniform_real_distribution<double> U(0.0, 1.0);
  for (auto& x : hv) x = U(gen);

  int64_t *dk;
  double *dv, *dsums;
  unsigned long long* drc;
  CHECK(hipMalloc(&dk, n * 8));
  CHECK(hipMalloc(&dv, n * 8));
  CHECK(hipMalloc(&dsums, keys * 8));
  CHECK(hipMalloc(&drc, keys * 8));
  CHECK(hipMemcpy(dv, hv.data(), n * 8, hipMemcpyHostToDevice));

  const int grid = 8192;
  for (int dist = 0; dist < 2; ++dist) {
    if (dist == 0) {
      std::uniform_int_distribution<int64_t> K(0, keys - 1);
      for (auto& x : hk) x = K(gen);
    } else {
      fill_zipf(hk, keys, 1.2, gen);
    }
    CHECK(hipMemcpy(dk, hk.data(), n * 8, hipMemcpyHostToDevice));
    // host reference on a prefix for correctness
    const int64_t chk = std::min<int64_t>(n, 2'000'000);
    std::vector<double> ref(keys, 0.0);
    for (int64_t i = 0; i < chk; ++i) ref[hk[i]] += hv[i];

    for (int var = 1; var <= 2; ++var) {
      auto run = [&](int64_t rows) {
        CHECK(hipMemset(dsums, 0, keys * 8));
        CHECK(hipMemset(drc, 0, keys * 8));
        hipEvent_t a, b;
        CHECK(hipEventCreate(&a));
        CHECK(hipEventCreate(&b));
        CHECK(hipEventRecord(a));
        if (var == 1)
          hipLaunchKernelGGL(k_v1, dim3(grid), dim3(BLOCK), 0, 0, dk, dv,
                             rows, dsums, drc);
        else
          hipLaunchKernelGGL(k_v2, dim3(grid), dim3(BLOCK), 0, 0, dk, dv,
                             rows, dsums, drc);
        CHECK(hipEventRecord(b));
        CHECK(hipEventSynchronize(b));
        float ms = 0;
        CHECK(hipEventElapsedTime(&ms, a, b));
        return ms;
      };
      // correctness on the prefix
      run(chk);
      std::vector<double> got(keys);
      CHECK(hipMemcpy(got.data(), dsums, keys * 8, hipMemcpyDeviceToHost));
      double maxrel = 0;
      for (int64_t s = 0; s < keys; ++s)
        if (ref[s] != 0.0)
          maxrel = std::max(maxrel,
                            std::abs(got[s] - ref[s]) /
                                std::max(1.0, std::abs(ref[s])));
      run(n);  // warm
      const float ms = run(n);
      printf("%-8s v%d: %8.2f ms  %6.1f G rows/s  (prefix maxrel %.2e)\n",
             dist == 0 ? "uniform" : "zipf1.2", var, ms, n / ms / 1e6,
             maxrel);
    }
  }
  return 0;
}
