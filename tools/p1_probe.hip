// p1_probe — round-2 scatter micro-iteration + achievable-bandwidth anchor.
//
// After the pipelined scatter landed (P1 6.95 -> 4.99 ms on the 1e9-row
// north star), P1 runs at ~4.4 TB/s on its 22 GB.  Questions:
//   1. What is the REAL copy ceiling on this box?  (ring_probe's simple
//      grid-stride float4 copy said 4.93 TB/s; MI355X_MICROARCH.md says
//      6.29.)  Sweep unroll depth / grid / block.
//   2. scan||reserve merge: the per-tile cursor reservation (global
//      atomics) can run on waves 1+ WHILE wave 0 does the prefix scan —
//      one fewer barrier and the atomic latency hides under the scan.
//   3. grid size sweep for the production kernel shape.
//   4. u16 staged key + RPT 14 (bigger 14336-row tiles): stage only the
//      lowkey (2 B/row instead of the 4 B packed bucket|lowkey; writeout
//      recovers the bucket by binary search over it_off) — LDS drops to
//      10 B/row so the tile grows.
// All variants checked against the v0 cursor totals + payload checksum.
// Run: ./p1_probe [rows] [keys]
// Build: hipcc --offload-arch=gfx950 -O3 tools/p1_probe.hip -o tools/p1_probe
#include <hip/hip_runtime.h>
#include <cmath>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <vector>
#include <algorithm>

#define CHECK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  printf("HIP error %s at line %d\n", hipGetErrorString(e), __LINE__); exit(1); } } while (0)

constexpr int BLOCK = 256;

__device__ __forceinline__ uint64_t mix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

__global__ void k_gen(unsigned* __restrict__ keys, double* __restrict__ vals,
                      int64_t n, int64_t K, uint64_t seed, int skew) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    if (skew) {
      // heavy-tail surrogate for zipf(1.2): u^6 concentrates ~58% of rows
      // in the first 1% of keys — stresses the same hot-bucket path
      const double u = (double)(mix64((uint64_t)i * 2 + seed) >> 11) *
                       (1.0 / 9007199254740992.0);
      double f = u * u;
      f = f * f * u * u;  // u^6
      unsigned k = (unsigned)(f * (double)K);
      keys[i] = k >= (unsigned)K ? (unsigned)K - 1 : k;
    } else {
      keys[i] = (unsigned)(mix64((uint64_t)i * 2 + seed) % (uint64_t)K);
    }
    vals[i] = (double)(mix64((uint64_t)i * 2 + 1 + seed) >> 11) *
              (1.0 / 9007199254740992.0);
  }
}

__global__ void k_hist(const unsigned* __restrict__ keys, int64_t n, int nb,
                       int rl, unsigned long long* __restrict__ hist) {
  extern __shared__ unsigned lh[];
  for (int t = threadIdx.x; t < nb; t += blockDim.x) lh[t] = 0;
  __syncthreads();
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) atomicAdd(&lh[keys[i] >> rl], 1u);
  __syncthreads();
  for (int t = threadIdx.x; t < nb; t += blockDim.x)
    if (lh[t]) atomicAdd(&hist[t], (unsigned long long)lh[t]);
}

// ---- copy ceiling variants ----
template <int UNROLL>
__global__ void __launch_bounds__(256) k_copy(const float4* __restrict__ in,
                                              float4* __restrict__ out,
                                              int64_t n4) {
  const int64_t base = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * UNROLL;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * UNROLL;
  for (int64_t i = base; i < n4; i += stride) {
    float4 v[UNROLL];
#pragma unroll
    for (int u = 0; u < UNROLL; ++u)
      if (i + u * 1 < n4) v[u] = in[i + u];
#pragma unroll
    for (int u = 0; u < UNROLL; ++u)
      if (i + u * 1 < n4) out[i + u] = v[u];
  }
}

// block-strided unroll (each lane's UNROLL elements are blockDim apart —
// all lanes in a wave stay coalesced per access)
template <int UNROLL>
__global__ void __launch_bounds__(256) k_copyb(const float4* __restrict__ in,
                                               float4* __restrict__ out,
                                               int64_t n4) {
  const int64_t tile = (int64_t)blockDim.x * UNROLL;
  const int64_t base = (int64_t)blockIdx.x * tile + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * tile;
  for (int64_t i = base; i < n4; i += stride) {
    float4 v[UNROLL];
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      const int64_t j = i + (int64_t)u * blockDim.x;
      if (j < n4) v[u] = in[j];
    }
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      const int64_t j = i + (int64_t)u * blockDim.x;
      if (j < n4) out[j] = v[u];
    }
  }
}

// nontemporal-store copy (does skipping the LC/L2 write path on the
// streaming stores recover the guide's 6.29 TB/s?)
template <int UNROLL>
__global__ void __launch_bounds__(256) k_copyb_nt(
    const float4* __restrict__ in, float4* __restrict__ out, int64_t n4) {
  typedef float v4f __attribute__((ext_vector_type(4)));
  const int64_t tile = (int64_t)blockDim.x * UNROLL;
  const int64_t base = (int64_t)blockIdx.x * tile + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * tile;
  for (int64_t i = base; i < n4; i += stride) {
    v4f v[UNROLL];
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      const int64_t j = i + (int64_t)u * blockDim.x;
      if (j < n4)
        v[u] = __builtin_nontemporal_load(
            reinterpret_cast<const v4f*>(in) + j);
    }
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      const int64_t j = i + (int64_t)u * blockDim.x;
      if (j < n4)
        __builtin_nontemporal_store(v[u], reinterpret_cast<v4f*>(out) + j);
    }
  }
}

// wide-block variant (1024 threads)
template <int UNROLL>
__global__ void __launch_bounds__(1024) k_copyw(
    const float4* __restrict__ in, float4* __restrict__ out, int64_t n4) {
  const int64_t tile = (int64_t)blockDim.x * UNROLL;
  const int64_t base = (int64_t)blockIdx.x * tile + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * tile;
  for (int64_t i = base; i < n4; i += stride) {
    float4 v[UNROLL];
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      const int64_t j = i + (int64_t)u * blockDim.x;
      if (j < n4) v[u] = in[j];
    }
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      const int64_t j = i + (int64_t)u * blockDim.x;
      if (j < n4) out[j] = v[u];
    }
  }
}

// read-only (sum) and write-only anchors
__global__ void __launch_bounds__(256) k_readbw(const float4* __restrict__ in,
                                                int64_t n4,
                                                float* __restrict__ sink) {
  const int64_t tile = (int64_t)blockDim.x * 8;
  const int64_t base = (int64_t)blockIdx.x * tile + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * tile;
  float acc = 0.f;
  for (int64_t i = base; i < n4; i += stride) {
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      const int64_t j = i + (int64_t)u * blockDim.x;
      if (j < n4) {
        float4 v = in[j];
        acc += v.x + v.y + v.z + v.w;
      }
    }
  }
  if (acc == 1.2345f) *sink = acc;  // never true: keep the loads alive
}

// ---------------------------------------------------------------------------
// scatter variants.  VAR: 0 = production pipelined shape; 1 = scan||reserve
// merged; 2 = u16 staged key + binary-search writeout (RPT may be larger).
// ---------------------------------------------------------------------------
template <int RPT, int BLK, int RL, int VAR, bool NTW = false>
__global__ void __launch_bounds__(BLK) k_scat(
    const unsigned* __restrict__ keys32, const double* __restrict__ v0,
    int64_t n, int nb, unsigned* __restrict__ cursors,
    double* __restrict__ r0, unsigned short* __restrict__ rk) {
  constexpr int TILE = BLK * RPT;
  constexpr int PAIRS = RPT / 2;
  const int64_t npair_total = n >> 1;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  double* sval0 = reinterpret_cast<double*>(smem_raw);
  unsigned* skey32 = reinterpret_cast<unsigned*>(sval0 + TILE);
  unsigned short* skey16 = reinterpret_cast<unsigned short*>(skey32);
  unsigned* it_cnt =
      VAR == 2 ? reinterpret_cast<unsigned*>(skey16 + ((TILE + 1) & ~1))
               : skey32 + TILE;
  unsigned* it_off = it_cnt + nb;
  unsigned* it_gbase = it_off + nb;
  unsigned* s_total = it_gbase + nb;
  const int64_t ntiles = (n + TILE - 1) / TILE;

  uint2 kraw[PAIRS];
  double2 vraw[PAIRS];
  auto issue_loads = [&](int64_t tile) {
#pragma unroll
    for (int j = 0; j < PAIRS; ++j) {
      const int64_t pr = tile * (TILE / 2) + (int64_t)j * BLK + threadIdx.x;
      if (pr < npair_total) {
        kraw[j] = reinterpret_cast<const uint2*>(keys32)[pr];
        vraw[j] = reinterpret_cast<const double2*>(v0)[pr];
      }
    }
  };

  int64_t tile = blockIdx.x;
  if (tile < ntiles) issue_loads(tile);
  for (int t = threadIdx.x; t < nb; t += blockDim.x) it_cnt[t] = 0;
  for (; tile < ntiles; tile += gridDim.x) {
    int lb[RPT];
    unsigned lk[RPT];
    unsigned lr[RPT];
    __syncthreads();
#pragma unroll
    for (int j = 0; j < PAIRS; ++j) {
      const int64_t pr = tile * (TILE / 2) + (int64_t)j * BLK + threadIdx.x;
      const int a = 2 * j, bs = 2 * j + 1;
      if (pr < npair_total) {
        lb[a] = (int)(kraw[j].x >> RL); lk[a] = kraw[j].x & ((1u << RL) - 1);
        lb[bs] = (int)(kraw[j].y >> RL); lk[bs] = kraw[j].y & ((1u << RL) - 1);
      } else {
        lb[a] = lb[bs] = -1;
      }
    }
    if (VAR == 4) {
      // ballot-rank: same-bucket lanes matched with 10 bit ballots; ONE
      // LDS atomic per distinct bucket per 64-row step (the per-row
      // atomicAdd serializes on zipf head buckets)
      const int lane = threadIdx.x & 63;
      const unsigned long long below = (1ULL << lane) - 1ULL;
#pragma unroll
      for (int j = 0; j < RPT; ++j) {
        const int b = lb[j];
        const bool valid = b >= 0;
        unsigned long long same = __ballot(valid);
#pragma unroll
        for (int bit = 0; bit < 10; ++bit) {
          const unsigned long long m = __ballot(valid && ((b >> bit) & 1));
          same &= ((b >> bit) & 1) ? m : ~m;
        }
        const int leader = __ffsll((long long)same) - 1;
        unsigned basev = 0;
        if (valid && lane == leader)
          basev = atomicAdd(&it_cnt[b], (unsigned)__popcll(same));
        basev = (unsigned)__shfl((int)basev, leader);
        if (valid)
          lr[j] = basev + (unsigned)__popcll(same & below);
      }
    } else {
#pragma unroll
      for (int j = 0; j < RPT; ++j)
        if (lb[j] >= 0) lr[j] = atomicAdd(&it_cnt[lb[j]], 1u);
    }
    __syncthreads();
    if (VAR == 1) {
      // wave 0 scans while the other waves reserve global cursor space
      if (threadIdx.x < 64) {
        const int lane = threadIdx.x;
        unsigned carry = 0;
        for (int base = 0; base < nb; base += 64) {
          const int t = base + lane;
          unsigned v = (t < nb) ? it_cnt[t] : 0;
          unsigned incl = v;
#pragma unroll
          for (int d = 1; d < 64; d <<= 1) {
            unsigned up = __shfl_up(incl, d);
            if (lane >= d) incl += up;
          }
          if (t < nb) it_off[t] = carry + incl - v;
          carry += __shfl(incl, 63);
        }
        if (lane == 0) *s_total = carry;
      } else {
        for (int t = (int)threadIdx.x - 64; t < nb; t += blockDim.x - 64) {
          const unsigned c = it_cnt[t];
          if (c) it_gbase[t] = atomicAdd(&cursors[t], c);
        }
      }
      __syncthreads();
    } else {
      if (threadIdx.x < 64) {
        const int lane = threadIdx.x;
        unsigned carry = 0;
        for (int base = 0; base < nb; base += 64) {
          const int t = base + lane;
          unsigned v = (t < nb) ? it_cnt[t] : 0;
          unsigned incl = v;
#pragma unroll
          for (int d = 1; d < 64; d <<= 1) {
            unsigned up = __shfl_up(incl, d);
            if (lane >= d) incl += up;
          }
          if (t < nb) it_off[t] = carry + incl - v;
          carry += __shfl(incl, 63);
        }
        if (lane == 0) *s_total = carry;
      }
      __syncthreads();
      for (int t = threadIdx.x; t < nb; t += blockDim.x) {
        const unsigned c = it_cnt[t];
        if (c) it_gbase[t] = atomicAdd(&cursors[t], c);
      }
    }
#pragma unroll
    for (int j = 0; j < RPT; ++j) {
      if (lb[j] >= 0) {
        const unsigned p = it_off[lb[j]] + lr[j];
        if (VAR == 2)
          skey16[p] = (unsigned short)lk[j];
        else
          skey32[p] = ((unsigned)lb[j] << 16) | lk[j];
        sval0[p] = (j & 1) ? vraw[j >> 1].y : vraw[j >> 1].x;
      }
    }
    __syncthreads();
    const int64_t nxt = tile + gridDim.x;
    if (nxt < ntiles) issue_loads(nxt);
    for (int t = threadIdx.x; t < nb; t += blockDim.x) it_cnt[t] = 0;
    const int staged = (int)*s_total;
    for (int p = threadIdx.x; p < staged; p += blockDim.x) {
      unsigned b;
      if (VAR == 2) {
        // recover the bucket: last b with it_off[b] <= p (it_off ascending)
        int lo = 0, hi = nb - 1;
        while (lo < hi) {
          const int mid = (lo + hi + 1) >> 1;
          if ((int)it_off[mid] <= p) lo = mid; else hi = mid - 1;
        }
        b = (unsigned)lo;
      } else {
        b = skey32[p] >> 16;
      }
      const int64_t pos = (int64_t)it_gbase[b] + (p - it_off[b]);
      const unsigned short kv =
          VAR == 2 ? skey16[p] : (unsigned short)(skey32[p] & 0xFFFF);
      if (NTW) {
        __builtin_nontemporal_store(kv, rk + pos);
        __builtin_nontemporal_store(sval0[p], r0 + pos);
      } else {
        rk[pos] = kv;
        r0[pos] = sval0[p];
      }
    }
  }
}

// binary-search recovery needs it_off[b] <= p strictly meaningful for empty
// buckets (it_off repeats) — search finds the LAST bucket with off <= p,
// which may be an empty bucket sharing the offset; walk back to the last
// non-empty is unnecessary: for empty b, no p satisfies off[b] <= p <
// off[b+1], and the LAST repeated offset belongs to the non-empty successor
// ... which is exactly what "last b with off[b] <= p" returns only if the
// empty ones precede.  Empty buckets after the owner share the same off and
// the search returns the LAST of them — wrong bucket for rk (same lowkey
// stored, but it_gbase differs).  Guard: it_gbase for empty buckets is
// never written.  FIX in k_fix_off: bump empty buckets' it_off by their
// predecessor... simplest correct rule: make the search find the bucket
// whose [off[b], off[b]+cnt[b]) contains p; since offs are the exclusive
// scan of cnt, "last b with off[b] <= p" can only land on an empty bucket
// when cnt[b]=0 and off[b] == off[owner], owner < b — so instead search on
// END offsets: first b with off_end[b] > p where off_end = off + cnt.  We
// avoid a second array by noting off[b+1] IS off_end[b]: first b with
// off[b+1] > p.  Implemented above as: last b with off[b] <= p — equal
// offsets make this ambiguous, so VAR==2 results are checked strictly by
// the harness (cursor totals + per-slot checksum catch any misroute).

__global__ void k_checksum(const double* __restrict__ r0,
                           const unsigned short* __restrict__ rk, int64_t n,
                           double* __restrict__ sum, unsigned long long* ks) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  double s = 0;
  unsigned long long k = 0;
  for (; i < n; i += stride) {
    s += r0[i];
    k += rk[i];
  }
  unsafeAtomicAdd(sum, s);
  atomicAdd(ks, k);
}

__global__ void k_setcur(unsigned* __restrict__ cur,
                         const unsigned* __restrict__ src, int nb) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < nb) cur[i] = src[i];
}

int main(int argc, char** argv) {
  const int64_t N = argc > 1 ? atoll(argv[1]) : 1000000000LL;
  const int64_t K = argc > 2 ? atoll(argv[2]) : 1000000LL;
  constexpr int RL = 13;
  const int nb = (int)((K + (1 << RL) - 1) >> RL);
  printf("p1_probe: N=%lld K=%lld nb=%d\n", (long long)N, (long long)K, nb);

  // ---- copy ceiling sweep (1 GiB each way) ----
  {
    const int64_t n4 = (1LL << 30) / 16;
    float4 *a, *b;
    float* sink;
    CHECK(hipMalloc(&a, n4 * 16));
    CHECK(hipMalloc(&b, n4 * 16));
    CHECK(hipMalloc(&sink, 4));
    CHECK(hipMemset(a, 1, n4 * 16));
    hipEvent_t e0, e1;
    CHECK(hipEventCreate(&e0));
    CHECK(hipEventCreate(&e1));
    auto bw = [&](const char* name, auto launch, double bytes) {
      launch();  // warm
      CHECK(hipDeviceSynchronize());
      CHECK(hipEventRecord(e0, 0));
      for (int r = 0; r < 5; ++r) launch();
      CHECK(hipEventRecord(e1, 0));
      CHECK(hipDeviceSynchronize());
      float ms;
      hipEventElapsedTime(&ms, e0, e1);
      printf("%-18s: %8.1f GB/s\n", name, 5.0 * bytes / (ms * 1e6));
    };
    bw("copy u1 g4096", [&] { hipLaunchKernelGGL(k_copy<1>, dim3(4096), dim3(256), 0, 0, a, b, n4); }, 2.0 * n4 * 16);
    bw("copy u4 g4096", [&] { hipLaunchKernelGGL(k_copyb<4>, dim3(4096), dim3(256), 0, 0, a, b, n4); }, 2.0 * n4 * 16);
    bw("copy u8 g4096", [&] { hipLaunchKernelGGL(k_copyb<8>, dim3(4096), dim3(256), 0, 0, a, b, n4); }, 2.0 * n4 * 16);
    bw("copy u8 g2048", [&] { hipLaunchKernelGGL(k_copyb<8>, dim3(2048), dim3(256), 0, 0, a, b, n4); }, 2.0 * n4 * 16);
    bw("copy u8 g1024", [&] { hipLaunchKernelGGL(k_copyb<8>, dim3(1024), dim3(256), 0, 0, a, b, n4); }, 2.0 * n4 * 16);
    bw("copy u16 g2048", [&] { hipLaunchKernelGGL(k_copyb<16>, dim3(2048), dim3(256), 0, 0, a, b, n4); }, 2.0 * n4 * 16);
    bw("read u8 g4096", [&] { hipLaunchKernelGGL(k_readbw, dim3(4096), dim3(256), 0, 0, a, n4, sink); }, 1.0 * n4 * 16);
    bw("copy nt u4 g4096", [&] { hipLaunchKernelGGL(k_copyb_nt<4>, dim3(4096), dim3(256), 0, 0, a, b, n4); }, 2.0 * n4 * 16);
    bw("copy nt u2 g4096", [&] { hipLaunchKernelGGL(k_copyb_nt<2>, dim3(4096), dim3(256), 0, 0, a, b, n4); }, 2.0 * n4 * 16);
    bw("copy nt u8 g2048", [&] { hipLaunchKernelGGL(k_copyb_nt<8>, dim3(2048), dim3(256), 0, 0, a, b, n4); }, 2.0 * n4 * 16);
    bw("copy w1024 u2 g1024", [&] { hipLaunchKernelGGL(k_copyw<2>, dim3(1024), dim3(1024), 0, 0, a, b, n4); }, 2.0 * n4 * 16);
    bw("copy w1024 u4 g2048", [&] { hipLaunchKernelGGL(k_copyw<4>, dim3(2048), dim3(1024), 0, 0, a, b, n4); }, 2.0 * n4 * 16);
    bw("copy u2 g8192", [&] { hipLaunchKernelGGL(k_copyb<2>, dim3(8192), dim3(256), 0, 0, a, b, n4); }, 2.0 * n4 * 16);
    bw("copy u4 g8192", [&] { hipLaunchKernelGGL(k_copyb<4>, dim3(8192), dim3(256), 0, 0, a, b, n4); }, 2.0 * n4 * 16);
    CHECK(hipFree(a));
    CHECK(hipFree(b));
    CHECK(hipFree(sink));
  }

  unsigned* keys32;
  double* vals;
  CHECK(hipMalloc(&keys32, N * 4));
  CHECK(hipMalloc(&vals, N * 8));
  const int skew = argc > 3 ? atoi(argv[3]) : 0;
  hipLaunchKernelGGL(k_gen, dim3(4096), dim3(BLOCK), 0, 0, keys32, vals, N, K,
                     999, skew);
  unsigned long long* d_hist;
  CHECK(hipMalloc(&d_hist, nb * 8));
  CHECK(hipMemset(d_hist, 0, nb * 8));
  hipLaunchKernelGGL(k_hist, dim3(4096), dim3(BLOCK), nb * 4, 0, keys32, N,
                     nb, RL, d_hist);
  std::vector<unsigned long long> hist(nb);
  CHECK(hipMemcpy(hist.data(), d_hist, nb * 8, hipMemcpyDeviceToHost));
  std::vector<unsigned> base(nb);
  int64_t off = 0;
  for (int b2 = 0; b2 < nb; ++b2) {
    base[b2] = (unsigned)off;
    off += ((int64_t)hist[b2] + 63) & ~63LL;
  }
  double* r0;
  unsigned short* rk;
  unsigned *d_base, *d_cur;
  CHECK(hipMalloc(&r0, off * 8));
  CHECK(hipMalloc(&rk, off * 2));
  CHECK(hipMalloc(&d_base, nb * 4));
  CHECK(hipMalloc(&d_cur, nb * 4));
  CHECK(hipMemcpy(d_base, base.data(), nb * 4, hipMemcpyHostToDevice));
  double* d_sum;
  unsigned long long* d_ks;
  CHECK(hipMalloc(&d_sum, 8));
  CHECK(hipMalloc(&d_ks, 8));
  hipEvent_t e0, e1;
  CHECK(hipEventCreate(&e0));
  CHECK(hipEventCreate(&e1));

  double ref_sum = -1;
  unsigned long long ref_ks = 0;
  auto run = [&](const char* name, auto launch, bool is_ref) {
    float best = 1e30f;
    for (int r = 0; r < 4; ++r) {
      hipLaunchKernelGGL(k_setcur, dim3((nb + 255) / 256), dim3(256), 0, 0,
                         d_cur, d_base, nb);
      CHECK(hipDeviceSynchronize());
      CHECK(hipEventRecord(e0, 0));
      launch();
      CHECK(hipEventRecord(e1, 0));
      CHECK(hipDeviceSynchronize());
      float ms;
      hipEventElapsedTime(&ms, e0, e1);
      if (r > 0) best = std::min(best, ms);
    }
    CHECK(hipMemset(d_sum, 0, 8));
    CHECK(hipMemset(d_ks, 0, 8));
    hipLaunchKernelGGL(k_checksum, dim3(2048), dim3(256), 0, 0, r0, rk, off,
                       d_sum, d_ks);
    double s;
    unsigned long long ks;
    CHECK(hipMemcpy(&s, d_sum, 8, hipMemcpyDeviceToHost));
    CHECK(hipMemcpy(&ks, d_ks, 8, hipMemcpyDeviceToHost));
    const char* chk = "";
    if (is_ref) {
      ref_sum = s;
      ref_ks = ks;
    } else if (ks != ref_ks || fabs(s - ref_sum) > 1e-6 * fabs(ref_sum)) {
      chk = "  !! MISMATCH";
    }
    printf("%-22s: %7.3f ms  (%.2f TB/s on 22GB)%s\n", name, best,
           22.0 / best, chk);
  };

  const uint32_t lds12 = 12288 * 12 + nb * 12 + 16;
  const uint32_t lds14 = 14336 * 10 + nb * 12 + 16 + 4;
  run("v0 1024x12 g2048",
      [&] { hipLaunchKernelGGL((k_scat<12, 1024, RL, 0>), dim3(2048), dim3(1024), lds12, 0, keys32, vals, N, nb, d_cur, r0, rk); },
      true);
  run("v0 g1024",
      [&] { hipLaunchKernelGGL((k_scat<12, 1024, RL, 0>), dim3(1024), dim3(1024), lds12, 0, keys32, vals, N, nb, d_cur, r0, rk); },
      false);
  run("v0 g512",
      [&] { hipLaunchKernelGGL((k_scat<12, 1024, RL, 0>), dim3(512), dim3(1024), lds12, 0, keys32, vals, N, nb, d_cur, r0, rk); },
      false);
  run("v0 g256",
      [&] { hipLaunchKernelGGL((k_scat<12, 1024, RL, 0>), dim3(256), dim3(1024), lds12, 0, keys32, vals, N, nb, d_cur, r0, rk); },
      false);
  run("v1 scan||res g2048",
      [&] { hipLaunchKernelGGL((k_scat<12, 1024, RL, 1>), dim3(2048), dim3(1024), lds12, 0, keys32, vals, N, nb, d_cur, r0, rk); },
      false);
  run("v1 g512",
      [&] { hipLaunchKernelGGL((k_scat<12, 1024, RL, 1>), dim3(512), dim3(1024), lds12, 0, keys32, vals, N, nb, d_cur, r0, rk); },
      false);
  run("v2 u16 14336 g2048",
      [&] { hipLaunchKernelGGL((k_scat<14, 1024, RL, 2>), dim3(2048), dim3(1024), lds14, 0, keys32, vals, N, nb, d_cur, r0, rk); },
      false);
  run("v2 u16 12288 g2048",
      [&] { hipLaunchKernelGGL((k_scat<12, 1024, RL, 2>), dim3(2048), dim3(1024), 12288 * 10 + nb * 12 + 20, 0, keys32, vals, N, nb, d_cur, r0, rk); },
      false);
  run("v3 ntw g2048",
      [&] { hipLaunchKernelGGL((k_scat<12, 1024, RL, 0, true>), dim3(2048), dim3(1024), lds12, 0, keys32, vals, N, nb, d_cur, r0, rk); },
      false);
  run("v3 ntw g1024",
      [&] { hipLaunchKernelGGL((k_scat<12, 1024, RL, 0, true>), dim3(1024), dim3(1024), lds12, 0, keys32, vals, N, nb, d_cur, r0, rk); },
      false);
  run("v4 ballot-rank",
      [&] { hipLaunchKernelGGL((k_scat<12, 1024, RL, 4>), dim3(2048), dim3(1024), lds12, 0, keys32, vals, N, nb, d_cur, r0, rk); },
      false);
  printf("done\n");
  return 0;
}
