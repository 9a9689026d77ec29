// radix_probe — variant sweep for the radix-groupby scatter/aggregate
// kernels (north-star shape: 1e9 rows, 1e6 keys).  Winners get folded into
// modin_amd/csrc/hipframe.hip.
// Build: hipcc --offload-arch=gfx950 -O3 tools/radix_probe.hip -o tools/radix_probe
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>
#include <cstring>
#include <functional>
#include <vector>
#include <random>
#include <algorithm>

#define CHECK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); exit(1); } } while (0)

constexpr int BLOCK = 256;

struct Work { int64_t start; int32_t bucket; int32_t len; };

// ---------------- scatter variants ----------------

// A: register-staged (current production shape), params RPT/RL
template <int RPT, int RL>
__global__ void __launch_bounds__(BLOCK) k_scat_reg(
    const int64_t* __restrict__ keys, const double* __restrict__ v0, int64_t n,
    int64_t n_slots, int nb, unsigned* __restrict__ cursors,
    double* __restrict__ r0, unsigned short* __restrict__ rk) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  unsigned* it_cnt = reinterpret_cast<unsigned*>(smem_raw);
  unsigned* it_base = it_cnt + nb;
  const int64_t TILE = (int64_t)BLOCK * RPT;
  const int64_t ntiles = (n + TILE - 1) / TILE;
  for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    const int64_t t0 = tile * TILE;
    for (int t = threadIdx.x; t < nb; t += blockDim.x) it_cnt[t] = 0;
    int lb[RPT];
    unsigned short lk[RPT];
    unsigned lr[RPT];
    double lv[RPT];
    __syncthreads();
#pragma unroll
    for (int j = 0; j < RPT; ++j) {
      const int64_t row = t0 + (int64_t)j * blockDim.x + threadIdx.x;
      lb[j] = -1;
      if (row < n) {
        const int64_t k = keys[row];
        lb[j] = (int)(k >> RL);
        lk[j] = (unsigned short)(k & ((1 << RL) - 1));
        lv[j] = v0[row];
      }
    }
#pragma unroll
    for (int j = 0; j < RPT; ++j)
      if (lb[j] >= 0) lr[j] = atomicAdd(&it_cnt[lb[j]], 1u);
    __syncthreads();
    for (int t = threadIdx.x; t < nb; t += blockDim.x) {
      const unsigned c = it_cnt[t];
      if (c) it_base[t] = atomicAdd(&cursors[t], c);
    }
    __syncthreads();
#pragma unroll
    for (int j = 0; j < RPT; ++j) {
      if (lb[j] >= 0) {
        const int64_t pos = (int64_t)it_base[lb[j]] + lr[j];
        rk[pos] = lk[j];
        r0[pos] = lv[j];
      }
    }
    __syncthreads();
  }
}

// B2: LDS-staged with wave-parallel exclusive scan of it_cnt
template <int RPT, int RL>
__global__ void __launch_bounds__(BLOCK) k_scat_lds2(
    const int64_t* __restrict__ keys, const double* __restrict__ v0, int64_t n,
    int64_t n_slots, int nb, unsigned* __restrict__ cursors,
    double* __restrict__ r0, unsigned short* __restrict__ rk) {
  constexpr int TILE = BLOCK * RPT;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  double* sval = reinterpret_cast<double*>(smem_raw);
  unsigned* skey = reinterpret_cast<unsigned*>(sval + TILE);
  unsigned* it_cnt = skey + TILE;
  unsigned* it_off = it_cnt + nb;
  unsigned* it_gbase = it_off + nb;
  const int64_t ntiles = (n + TILE - 1) / TILE;
  for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    const int64_t t0 = tile * TILE;
    for (int t = threadIdx.x; t < nb; t += blockDim.x) it_cnt[t] = 0;
    int lb[RPT];
    unsigned lk[RPT];
    unsigned lr[RPT];
    double lv[RPT];
    __syncthreads();
#pragma unroll
    for (int j = 0; j < RPT; ++j) {
      const int64_t row = t0 + (int64_t)j * blockDim.x + threadIdx.x;
      lb[j] = -1;
      if (row < n) {
        const int64_t k = keys[row];
        lb[j] = (int)(k >> RL);
        lk[j] = (unsigned)(k & ((1 << RL) - 1));
        lv[j] = v0[row];
      }
    }
#pragma unroll
    for (int j = 0; j < RPT; ++j)
      if (lb[j] >= 0) lr[j] = atomicAdd(&it_cnt[lb[j]], 1u);
    __syncthreads();
    // wave 0: parallel exclusive scan over nb counters (chunks of 64 lanes)
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
    }
    __syncthreads();
    for (int t = threadIdx.x; t < nb; t += blockDim.x) {
      const unsigned c = it_cnt[t];
      if (c) it_gbase[t] = atomicAdd(&cursors[t], c);
    }
#pragma unroll
    for (int j = 0; j < RPT; ++j) {
      if (lb[j] >= 0) {
        const unsigned p = it_off[lb[j]] + lr[j];
        sval[p] = lv[j];
        skey[p] = ((unsigned)lb[j] << 16) | lk[j];
      }
    }
    __syncthreads();
    const int valid = (int)min((int64_t)TILE, n - t0);
    for (int p = threadIdx.x; p < valid; p += blockDim.x) {
      const unsigned b = skey[p] >> 16;
      const int64_t pos = (int64_t)it_gbase[b] + (p - it_off[b]);
      r0[pos] = sval[p];
      rk[pos] = (unsigned short)(skey[p] & 0xFFFF);
    }
    __syncthreads();
  }
}

// B3: like B2 but 16 B vectorized loads (lane owns consecutive row PAIRS)
// and double2 LDS reads in the writeout sweep.
template <int RPT, int RL, int BLK>
__global__ void __launch_bounds__(BLK) k_scat_lds3(
    const int64_t* __restrict__ keys, const double* __restrict__ v0, int64_t n,
    int64_t n_slots, int nb, unsigned* __restrict__ cursors,
    double* __restrict__ r0, unsigned short* __restrict__ rk) {
  constexpr int TILE = BLK * RPT;
  constexpr int PAIRS = RPT / 2;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  double* sval = reinterpret_cast<double*>(smem_raw);
  unsigned* skey = reinterpret_cast<unsigned*>(sval + TILE);
  unsigned* it_cnt = skey + TILE;
  unsigned* it_off = it_cnt + nb;
  unsigned* it_gbase = it_off + nb;
  unsigned* s_total = it_gbase + nb;
  const int64_t ntiles = (n + TILE - 1) / TILE;
  const int64_t npair_total = n >> 1;
  for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    const int64_t t0 = tile * TILE;
    for (int t = threadIdx.x; t < nb; t += blockDim.x) it_cnt[t] = 0;
    int lb[RPT];
    unsigned lk[RPT];
    unsigned lr[RPT];
    double lv[RPT];
    __syncthreads();
#pragma unroll
    for (int j = 0; j < PAIRS; ++j) {
      // pair index within the whole array: tile base/2 + j*BLK + tid
      const int64_t pr = (t0 >> 1) + (int64_t)j * BLK + threadIdx.x;
      const int a = 2 * j, b = 2 * j + 1;
      lb[a] = lb[b] = -1;
      if (pr < npair_total) {
        const longlong2 kk = reinterpret_cast<const longlong2*>(keys)[pr];
        const double2 vv = reinterpret_cast<const double2*>(v0)[pr];
        lb[a] = (int)(kk.x >> RL);
        lk[a] = (unsigned)(kk.x & ((1 << RL) - 1));
        lv[a] = vv.x;
        lb[b] = (int)(kk.y >> RL);
        lk[b] = (unsigned)(kk.y & ((1 << RL) - 1));
        lv[b] = vv.y;
      }
    }
    // NOTE: odd-n tail handled by caller in production; probe n is even
#pragma unroll
    for (int j = 0; j < RPT; ++j)
      if (lb[j] >= 0) lr[j] = atomicAdd(&it_cnt[lb[j]], 1u);
    __syncthreads();
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
#pragma unroll
    for (int j = 0; j < RPT; ++j) {
      if (lb[j] >= 0) {
        const unsigned p = it_off[lb[j]] + lr[j];
        sval[p] = lv[j];
        skey[p] = ((unsigned)lb[j] << 16) | lk[j];
      }
    }
    __syncthreads();
    const int staged = (int)*s_total;
    for (int p = threadIdx.x; p < staged; p += blockDim.x) {
      const unsigned b = skey[p] >> 16;
      const int64_t pos = (int64_t)it_gbase[b] + (p - it_off[b]);
      r0[pos] = sval[p];
      rk[pos] = (unsigned short)(skey[p] & 0xFFFF);
    }
    __syncthreads();
  }
}

// B: LDS-staged bucket-sorted tile, coalesced writes.  TILE rows staged in
// LDS (val 8B + packed (bucket<<RL)|lowkey u32), written out in sorted order.
template <int RPT, int RL>
__global__ void __launch_bounds__(BLOCK) k_scat_lds(
    const int64_t* __restrict__ keys, const double* __restrict__ v0, int64_t n,
    int64_t n_slots, int nb, unsigned* __restrict__ cursors,
    double* __restrict__ r0, unsigned short* __restrict__ rk) {
  constexpr int TILE = BLOCK * RPT;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  double* sval = reinterpret_cast<double*>(smem_raw);                 // TILE*8
  unsigned* skey = reinterpret_cast<unsigned*>(sval + TILE);          // TILE*4
  unsigned* it_cnt = skey + TILE;                                     // nb
  unsigned* it_off = it_cnt + nb;                                     // nb (excl scan)
  unsigned* it_gbase = it_off + nb;                                   // nb
  const int64_t ntiles = (n + TILE - 1) / TILE;
  for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    const int64_t t0 = tile * TILE;
    for (int t = threadIdx.x; t < nb; t += blockDim.x) it_cnt[t] = 0;
    int lb[RPT];
    unsigned lk[RPT];
    unsigned lr[RPT];
    double lv[RPT];
    __syncthreads();
#pragma unroll
    for (int j = 0; j < RPT; ++j) {
      const int64_t row = t0 + (int64_t)j * blockDim.x + threadIdx.x;
      lb[j] = -1;
      if (row < n) {
        const int64_t k = keys[row];
        lb[j] = (int)(k >> RL);
        lk[j] = (unsigned)(k & ((1 << RL) - 1));
        lv[j] = v0[row];
      }
    }
#pragma unroll
    for (int j = 0; j < RPT; ++j)
      if (lb[j] >= 0) lr[j] = atomicAdd(&it_cnt[lb[j]], 1u);
    __syncthreads();
    // exclusive scan of it_cnt (single lane; nb <= 1024 — cheap vs the tile)
    if (threadIdx.x == 0) {
      unsigned run = 0;
      for (int t = 0; t < nb; ++t) { it_off[t] = run; run += it_cnt[t]; }
    }
    __syncthreads();
    for (int t = threadIdx.x; t < nb; t += blockDim.x) {
      const unsigned c = it_cnt[t];
      if (c) it_gbase[t] = atomicAdd(&cursors[t], c);
    }
    // stage bucket-sorted
#pragma unroll
    for (int j = 0; j < RPT; ++j) {
      if (lb[j] >= 0) {
        const unsigned p = it_off[lb[j]] + lr[j];
        sval[p] = lv[j];
        skey[p] = ((unsigned)lb[j] << 16) | lk[j];
      }
    }
    __syncthreads();
    // write out in sorted order: consecutive threads -> consecutive global
    // positions within each bucket segment
    const int valid = (int)min((int64_t)TILE, n - t0);
    for (int p = threadIdx.x; p < valid; p += blockDim.x) {
      const unsigned b = skey[p] >> 16;
      const int64_t pos = (int64_t)it_gbase[b] + (p - it_off[b]);
      r0[pos] = sval[p];
      rk[pos] = (unsigned short)(skey[p] & 0xFFFF);
    }
    __syncthreads();
  }
}

// B4: lds3 + (a) scan by wave0 WHILE other waves reserve cursors, (b)
// paired writeout (2 entries/lane: b128 LDS val reads, fused stores when the
// pair shares a bucket).
template <int RPT, int RL, int BLK>
__global__ void __launch_bounds__(BLK) k_scat_lds4(
    const int64_t* __restrict__ keys, const double* __restrict__ v0, int64_t n,
    int64_t n_slots, int nb, unsigned* __restrict__ cursors,
    double* __restrict__ r0, unsigned short* __restrict__ rk) {
  constexpr int TILE = BLK * RPT;
  constexpr int PAIRS = RPT / 2;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  double* sval = reinterpret_cast<double*>(smem_raw);
  unsigned* skey = reinterpret_cast<unsigned*>(sval + TILE);
  unsigned* it_cnt = skey + TILE;
  unsigned* it_off = it_cnt + nb;
  unsigned* it_gbase = it_off + nb;
  unsigned* s_total = it_gbase + nb;
  const int64_t ntiles = (n + TILE - 1) / TILE;
  const int64_t npair_total = n >> 1;
  for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    const int64_t t0 = tile * TILE;
    for (int t = threadIdx.x; t < nb; t += blockDim.x) it_cnt[t] = 0;
    int lb[RPT];
    unsigned lk[RPT];
    unsigned lr[RPT];
    double lv[RPT];
    __syncthreads();
#pragma unroll
    for (int j = 0; j < PAIRS; ++j) {
      const int64_t pr = (t0 >> 1) + (int64_t)j * BLK + threadIdx.x;
      const int a = 2 * j, b = 2 * j + 1;
      lb[a] = lb[b] = -1;
      if (pr < npair_total) {
        const longlong2 kk = reinterpret_cast<const longlong2*>(keys)[pr];
        const double2 vv = reinterpret_cast<const double2*>(v0)[pr];
        lb[a] = (int)(kk.x >> RL);
        lk[a] = (unsigned)(kk.x & ((1 << RL) - 1));
        lv[a] = vv.x;
        lb[b] = (int)(kk.y >> RL);
        lk[b] = (unsigned)(kk.y & ((1 << RL) - 1));
        lv[b] = vv.y;
      }
    }
#pragma unroll
    for (int j = 0; j < RPT; ++j)
      if (lb[j] >= 0) lr[j] = atomicAdd(&it_cnt[lb[j]], 1u);
    __syncthreads();
    // wave 0 scans; the OTHER waves reserve global cursors concurrently
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
#pragma unroll
    for (int j = 0; j < RPT; ++j) {
      if (lb[j] >= 0) {
        const unsigned p = it_off[lb[j]] + lr[j];
        sval[p] = lv[j];
        skey[p] = ((unsigned)lb[j] << 16) | lk[j];
      }
    }
    __syncthreads();
    // paired writeout
    const int staged = (int)*s_total;
    const int half = (staged + 1) >> 1;
    for (int q = threadIdx.x; q < half; q += blockDim.x) {
      const int p = 2 * q;
      const unsigned ka = skey[p];
      const unsigned ba = ka >> 16;
      const int64_t pa = (int64_t)it_gbase[ba] + (p - it_off[ba]);
      if (p + 1 < staged) {
        const unsigned kb2 = skey[p + 1];
        const unsigned bb = kb2 >> 16;
        const double2 vv = *reinterpret_cast<const double2*>(&sval[p]);
        if (bb == ba) {
          r0[pa] = vv.x;
          r0[pa + 1] = vv.y;
          const unsigned pack = (ka & 0xFFFF) | ((kb2 & 0xFFFF) << 16);
          if ((pa & 1) == 0)
            *reinterpret_cast<unsigned*>(&rk[pa]) = pack;
          else {
            rk[pa] = (unsigned short)(ka & 0xFFFF);
            rk[pa + 1] = (unsigned short)(kb2 & 0xFFFF);
          }
        } else {
          const int64_t pb = (int64_t)it_gbase[bb] + (p + 1 - it_off[bb]);
          r0[pa] = vv.x;
          r0[pb] = vv.y;
          rk[pa] = (unsigned short)(ka & 0xFFFF);
          rk[pb] = (unsigned short)(kb2 & 0xFFFF);
        }
      } else {
        r0[pa] = sval[p];
        rk[pa] = (unsigned short)(ka & 0xFFFF);
      }
    }
    __syncthreads();
  }
}

// B5: lds3 with SPLIT loads — keys load + ranks first, value loads issued
// AFTER ranks (latency hides under scan/reserve/barriers; plain loads are
// not drained by __syncthreads).
template <int RPT, int RL, int BLK>
__global__ void __launch_bounds__(BLK) k_scat_lds5(
    const int64_t* __restrict__ keys, const double* __restrict__ v0, int64_t n,
    int64_t n_slots, int nb, unsigned* __restrict__ cursors,
    double* __restrict__ r0, unsigned short* __restrict__ rk) {
  constexpr int TILE = BLK * RPT;
  constexpr int PAIRS = RPT / 2;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  double* sval = reinterpret_cast<double*>(smem_raw);
  unsigned* skey = reinterpret_cast<unsigned*>(sval + TILE);
  unsigned* it_cnt = skey + TILE;
  unsigned* it_off = it_cnt + nb;
  unsigned* it_gbase = it_off + nb;
  unsigned* s_total = it_gbase + nb;
  const int64_t ntiles = (n + TILE - 1) / TILE;
  const int64_t npair_total = n >> 1;
  for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    const int64_t t0 = tile * TILE;
    for (int t = threadIdx.x; t < nb; t += blockDim.x) it_cnt[t] = 0;
    int lb[RPT];
    unsigned lk[RPT];
    unsigned lr[RPT];
    double lv[RPT];
    __syncthreads();
#pragma unroll
    for (int j = 0; j < PAIRS; ++j) {
      const int64_t pr = (t0 >> 1) + (int64_t)j * BLK + threadIdx.x;
      const int a = 2 * j, b = 2 * j + 1;
      lb[a] = lb[b] = -1;
      if (pr < npair_total) {
        const longlong2 kk = reinterpret_cast<const longlong2*>(keys)[pr];
        lb[a] = (int)(kk.x >> RL);
        lk[a] = (unsigned)(kk.x & ((1 << RL) - 1));
        lb[b] = (int)(kk.y >> RL);
        lk[b] = (unsigned)(kk.y & ((1 << RL) - 1));
      }
    }
#pragma unroll
    for (int j = 0; j < RPT; ++j)
      if (lb[j] >= 0) lr[j] = atomicAdd(&it_cnt[lb[j]], 1u);
    // value loads issued NOW: latency hides under scan + reserve + barriers
#pragma unroll
    for (int j = 0; j < PAIRS; ++j) {
      const int64_t pr = (t0 >> 1) + (int64_t)j * BLK + threadIdx.x;
      if (pr < npair_total) {
        const double2 vv = reinterpret_cast<const double2*>(v0)[pr];
        lv[2 * j] = vv.x;
        lv[2 * j + 1] = vv.y;
      }
    }
    __syncthreads();
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
#pragma unroll
    for (int j = 0; j < RPT; ++j) {
      if (lb[j] >= 0) {
        const unsigned p = it_off[lb[j]] + lr[j];
        sval[p] = lv[j];
        skey[p] = ((unsigned)lb[j] << 16) | lk[j];
      }
    }
    __syncthreads();
    const int staged = (int)*s_total;
    for (int p = threadIdx.x; p < staged; p += blockDim.x) {
      const unsigned b = skey[p] >> 16;
      const int64_t pos = (int64_t)it_gbase[b] + (p - it_off[b]);
      r0[pos] = sval[p];
      rk[pos] = (unsigned short)(skey[p] & 0xFFFF);
    }
    __syncthreads();
  }
}

// B7: per-wave rank counters — it_cnt is [WPB][nb] so rank atomics never
// contend across waves; the scan folds wave offsets into per-(wave,bucket)
// bases.  WPB = BLK/64.
template <int RPT, int RL, int BLK>
__global__ void __launch_bounds__(BLK) k_scat_lds7(
    const int64_t* __restrict__ keys, const double* __restrict__ v0, int64_t n,
    int64_t n_slots, int nb, unsigned* __restrict__ cursors,
    double* __restrict__ r0, unsigned short* __restrict__ rk) {
  constexpr int TILE = BLK * RPT;
  constexpr int PAIRS = RPT / 2;
  constexpr int WPB = BLK / 64;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  double* sval = reinterpret_cast<double*>(smem_raw);
  unsigned* skey = reinterpret_cast<unsigned*>(sval + TILE);
  unsigned* it_cnt = skey + TILE;        // [WPB][nb]
  unsigned* it_base = it_cnt + WPB * nb; // [WPB][nb] per-(wave,bucket) base
  unsigned* it_off = it_base + WPB * nb; // [nb] block-level offsets
  unsigned* it_gbase = it_off + nb;      // [nb]
  unsigned* s_total = it_gbase + nb;
  const int wave = threadIdx.x >> 6;
  const int64_t ntiles = (n + TILE - 1) / TILE;
  const int64_t npair_total = n >> 1;
  for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    const int64_t t0 = tile * TILE;
    for (int t = threadIdx.x; t < WPB * nb; t += blockDim.x) it_cnt[t] = 0;
    int lb[RPT];
    unsigned lk[RPT];
    unsigned lr[RPT];
    double lv[RPT];
    __syncthreads();
#pragma unroll
    for (int j = 0; j < PAIRS; ++j) {
      const int64_t pr = (t0 >> 1) + (int64_t)j * BLK + threadIdx.x;
      const int a = 2 * j, b = 2 * j + 1;
      lb[a] = lb[b] = -1;
      if (pr < npair_total) {
        const longlong2 kk = reinterpret_cast<const longlong2*>(keys)[pr];
        const double2 vv = reinterpret_cast<const double2*>(v0)[pr];
        lb[a] = (int)(kk.x >> RL);
        lk[a] = (unsigned)(kk.x & ((1 << RL) - 1));
        lv[a] = vv.x;
        lb[b] = (int)(kk.y >> RL);
        lk[b] = (unsigned)(kk.y & ((1 << RL) - 1));
        lv[b] = vv.y;
      }
    }
    unsigned* mycnt = it_cnt + wave * nb;
#pragma unroll
    for (int j = 0; j < RPT; ++j)
      if (lb[j] >= 0) lr[j] = atomicAdd(&mycnt[lb[j]], 1u);
    __syncthreads();
    // wave 0: per-bucket totals + block prefix + per-wave bases
    if (threadIdx.x < 64) {
      const int lane = threadIdx.x;
      unsigned carry = 0;
      for (int base = 0; base < nb; base += 64) {
        const int t = base + lane;
        unsigned tot = 0, wsum[WPB];
        if (t < nb) {
#pragma unroll
          for (int w = 0; w < WPB; ++w) {
            wsum[w] = it_cnt[w * nb + t];
            tot += wsum[w];
          }
        }
        unsigned incl = tot;
#pragma unroll
        for (int d = 1; d < 64; d <<= 1) {
          unsigned up = __shfl_up(incl, d);
          if (lane >= d) incl += up;
        }
        if (t < nb) {
          unsigned off = carry + incl - tot;
          it_off[t] = off;
#pragma unroll
          for (int w = 0; w < WPB; ++w) {
            it_base[w * nb + t] = off;
            off += wsum[w];
          }
        }
        carry += __shfl(incl, 63);
      }
      if (lane == 0) *s_total = carry;
    }
    __syncthreads();
    for (int t = threadIdx.x; t < nb; t += blockDim.x) {
      unsigned c = 0;
#pragma unroll
      for (int w = 0; w < WPB; ++w) c += it_cnt[w * nb + t];
      if (c) it_gbase[t] = atomicAdd(&cursors[t], c);
    }
    unsigned* mybase = it_base + wave * nb;
#pragma unroll
    for (int j = 0; j < RPT; ++j) {
      if (lb[j] >= 0) {
        const unsigned p = mybase[lb[j]] + lr[j];
        sval[p] = lv[j];
        skey[p] = ((unsigned)lb[j] << 16) | lk[j];
      }
    }
    __syncthreads();
    const int staged = (int)*s_total;
    for (int p = threadIdx.x; p < staged; p += blockDim.x) {
      const unsigned b = skey[p] >> 16;
      const int64_t pos = (int64_t)it_gbase[b] + (p - it_off[b]);
      r0[pos] = sval[p];
      rk[pos] = (unsigned short)(skey[p] & 0xFFFF);
    }
    __syncthreads();
  }
}

// B8: wave-autonomous tiles — one WAVE owns each tile; ranks/scan/reserve/
// stage/writeout all wave-local, zero block barriers (waves self-overlap).
template <int RPT, int RL, int BLK>
__global__ void __launch_bounds__(BLK) k_scat_wave(
    const int64_t* __restrict__ keys, const double* __restrict__ v0, int64_t n,
    int64_t n_slots, int nb, unsigned* __restrict__ cursors,
    double* __restrict__ r0, unsigned short* __restrict__ rk) {
  constexpr int WPB = BLK / 64;
  constexpr int TILE = 64 * RPT;          // rows per wave-tile
  constexpr int PAIRS = RPT / 2;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  // per-wave carve: sval[TILE] f64 + skey[TILE] u32 + cnt/off/gbase[nb each]
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  char* my = smem_raw + (size_t)wave * (TILE * 12 + 3 * nb * 4 + 16);
  double* sval = reinterpret_cast<double*>(my);
  unsigned* skey = reinterpret_cast<unsigned*>(sval + TILE);
  unsigned* it_cnt = skey + TILE;
  unsigned* it_off = it_cnt + nb;
  unsigned* it_gbase = it_off + nb;
  const int64_t ntiles = (n + TILE - 1) / TILE;
  const int64_t npair_total = n >> 1;
  for (int64_t tile = (int64_t)blockIdx.x * WPB + wave; tile < ntiles;
       tile += (int64_t)gridDim.x * WPB) {
    const int64_t t0 = tile * TILE;
    for (int t = lane; t < nb; t += 64) it_cnt[t] = 0;
    int lb[RPT];
    unsigned lk[RPT];
    unsigned lr[RPT];
    double lv[RPT];
#pragma unroll
    for (int j = 0; j < PAIRS; ++j) {
      const int64_t pr = (t0 >> 1) + (int64_t)j * 64 + lane;
      const int a = 2 * j, b = 2 * j + 1;
      lb[a] = lb[b] = -1;
      if (pr < npair_total) {
        const longlong2 kk = reinterpret_cast<const longlong2*>(keys)[pr];
        const double2 vv = reinterpret_cast<const double2*>(v0)[pr];
        lb[a] = (int)(kk.x >> RL);
        lk[a] = (unsigned)(kk.x & ((1 << RL) - 1));
        lv[a] = vv.x;
        lb[b] = (int)(kk.y >> RL);
        lk[b] = (unsigned)(kk.y & ((1 << RL) - 1));
        lv[b] = vv.y;
      }
    }
#pragma unroll
    for (int j = 0; j < RPT; ++j)
      if (lb[j] >= 0) lr[j] = atomicAdd(&it_cnt[lb[j]], 1u);
    // wave-local exclusive scan of nb counters
    {
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
    }
    // wave reserve (lanes stride the buckets)
    for (int t = lane; t < nb; t += 64) {
      const unsigned c = it_cnt[t];
      if (c) it_gbase[t] = atomicAdd(&cursors[t], c);
    }
    // stage bucket-sorted
#pragma unroll
    for (int j = 0; j < RPT; ++j) {
      if (lb[j] >= 0) {
        const unsigned p = it_off[lb[j]] + lr[j];
        sval[p] = lv[j];
        skey[p] = ((unsigned)lb[j] << 16) | lk[j];
      }
    }
    // writeout (wave-lockstep; LDS deps handled by lgkmcnt)
    const int staged = (int)min((int64_t)TILE, n - t0);
    for (int p = lane; p < staged; p += 64) {
      const unsigned b = skey[p] >> 16;
      const int64_t pos = (int64_t)it_gbase[b] + (p - it_off[b]);
      r0[pos] = sval[p];
      rk[pos] = (unsigned short)(skey[p] & 0xFFFF);
    }
  }
}

// B6: ablation — template-disable phases of the lds3 structure to find the
// dominant cost.  PH bitmask: 1=ranks, 2=scan, 4=reserve, 8=stage, 16=writeout
template <int RPT, int RL, int BLK, int PH>
__global__ void __launch_bounds__(BLK) k_scat_abl(
    const int64_t* __restrict__ keys, const double* __restrict__ v0, int64_t n,
    int64_t n_slots, int nb, unsigned* __restrict__ cursors,
    double* __restrict__ r0, unsigned short* __restrict__ rk) {
  constexpr int TILE = BLK * RPT;
  constexpr int PAIRS = RPT / 2;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  double* sval = reinterpret_cast<double*>(smem_raw);
  unsigned* skey = reinterpret_cast<unsigned*>(sval + TILE);
  unsigned* it_cnt = skey + TILE;
  unsigned* it_off = it_cnt + nb;
  unsigned* it_gbase = it_off + nb;
  unsigned* s_total = it_gbase + nb;
  const int64_t ntiles = (n + TILE - 1) / TILE;
  const int64_t npair_total = n >> 1;
  for (int64_t tile = blockIdx.x; tile < ntiles; tile += gridDim.x) {
    const int64_t t0 = tile * TILE;
    for (int t = threadIdx.x; t < nb; t += blockDim.x) {
      it_cnt[t] = 0;
      if (!(PH & 4)) it_gbase[t] = (unsigned)(t * 64);  // fake bases
    }
    int lb[RPT];
    unsigned lk[RPT];
    unsigned lr[RPT];
    double lv[RPT];
    __syncthreads();
#pragma unroll
    for (int j = 0; j < PAIRS; ++j) {
      const int64_t pr = (t0 >> 1) + (int64_t)j * BLK + threadIdx.x;
      const int a = 2 * j, b = 2 * j + 1;
      lb[a] = lb[b] = -1;
      if (pr < npair_total) {
        const longlong2 kk = reinterpret_cast<const longlong2*>(keys)[pr];
        const double2 vv = reinterpret_cast<const double2*>(v0)[pr];
        lb[a] = (int)(kk.x >> RL);
        lk[a] = (unsigned)(kk.x & ((1 << RL) - 1));
        lv[a] = vv.x;
        lb[b] = (int)(kk.y >> RL);
        lk[b] = (unsigned)(kk.y & ((1 << RL) - 1));
        lv[b] = vv.y;
      }
    }
    if (PH & 1) {
#pragma unroll
      for (int j = 0; j < RPT; ++j)
        if (lb[j] >= 0) lr[j] = atomicAdd(&it_cnt[lb[j]], 1u);
    } else {
#pragma unroll
      for (int j = 0; j < RPT; ++j) lr[j] = (unsigned)(threadIdx.x + j) & 63;
    }
    __syncthreads();
    if (PH & 2) {
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
    } else if (threadIdx.x == 0) {
      *s_total = (unsigned)min((int64_t)TILE, n - t0);
    }
    __syncthreads();
    if (PH & 4) {
      for (int t = threadIdx.x; t < nb; t += blockDim.x) {
        const unsigned c = it_cnt[t];
        if (c) it_gbase[t] = atomicAdd(&cursors[t], c);
      }
    }
    if (PH & 8) {
#pragma unroll
      for (int j = 0; j < RPT; ++j) {
        if (lb[j] >= 0) {
          unsigned p = (PH & 2) ? it_off[lb[j]] + lr[j]
                                : (unsigned)(j * BLK + threadIdx.x);
          if (p >= TILE) p = TILE - 1;  // ablation safety
          sval[p] = lv[j];
          skey[p] = ((unsigned)lb[j] << 16) | lk[j];
        }
      }
    }
    __syncthreads();
    if (PH & 16) {
      const int staged = (int)*s_total;
      for (int p = threadIdx.x; p < staged; p += blockDim.x) {
        const unsigned b = skey[p] >> 16;
        int64_t pos = (int64_t)it_gbase[b] + (p - it_off[b]);
        pos &= (1LL << 30) - 1;  // ablation safety (alloc > 2^30 rows)
        r0[pos] = sval[p];
        rk[pos] = (unsigned short)(skey[p] & 0xFFFF);
      }
    }
    __syncthreads();
  }
}

// ---------------- aggregate variants ----------------

// presence-byte agg (production shape after the rowcnt->presence change)
template <int RL, int BLK>
__global__ void __launch_bounds__(BLK) k_agg_u8(
    const double* __restrict__ vals, const unsigned short* __restrict__ lowkeys,
    const Work* __restrict__ work, int64_t n_slots, double* __restrict__ gsums,
    unsigned long long* __restrict__ growcnt) {
  constexpr int RANGE = 1 << RL;
  __shared__ double lsums[RANGE];
  __shared__ unsigned char ltouch[RANGE];
  const Work w = work[blockIdx.x];
  for (int s = threadIdx.x; s < RANGE; s += blockDim.x) {
    lsums[s] = 0.0;
    ltouch[s] = 0;
  }
  __syncthreads();
  const int64_t npair = (int64_t)(w.len) >> 1;
  const ushort2* k2 = reinterpret_cast<const ushort2*>(lowkeys + w.start);
  const double2* v2 = reinterpret_cast<const double2*>(vals + w.start);
  for (int64_t i = threadIdx.x; i < npair; i += blockDim.x) {
    const ushort2 kk = k2[i];
    const double2 vv = v2[i];
    ltouch[kk.x] = 1;
    ltouch[kk.y] = 1;
    if (vv.x == vv.x) unsafeAtomicAdd(&lsums[kk.x], vv.x);
    if (vv.y == vv.y) unsafeAtomicAdd(&lsums[kk.y], vv.y);
  }
  __syncthreads();
  const int64_t gbase = (int64_t)w.bucket << RL;
  for (int s = threadIdx.x; s < RANGE; s += blockDim.x) {
    if (!ltouch[s] || gbase + s >= n_slots) continue;
    unsafeAtomicAdd(&gsums[gbase + s], lsums[s]);
    atomicAdd(&growcnt[gbase + s], 1ULL);
  }
}

template <int RL, int VEC, int BLK = 512>
__global__ void __launch_bounds__(BLK) k_agg(
    const double* __restrict__ vals, const unsigned short* __restrict__ lowkeys,
    const Work* __restrict__ work, int64_t n_slots, double* __restrict__ gsums,
    unsigned long long* __restrict__ growcnt) {
  constexpr int RANGE = 1 << RL;
  __shared__ double lsums[RANGE];
  __shared__ unsigned ltouch[RANGE];
  const Work w = work[blockIdx.x];
  for (int s = threadIdx.x; s < RANGE; s += blockDim.x) {
    lsums[s] = 0.0;
    ltouch[s] = 0;
  }
  __syncthreads();
  const int64_t end = w.start + w.len;
  if (VEC == 1) {
    for (int64_t i = w.start + threadIdx.x; i < end; i += blockDim.x) {
      const int slot = lowkeys[i];
      atomicAdd(&ltouch[slot], 1u);
      const double v = vals[i];
      if (v == v) unsafeAtomicAdd(&lsums[slot], v);
    }
  } else {
    // 2 rows per lane: ushort2 + double2 (start is 64-aligned)
    const int64_t npair = (int64_t)(w.len) >> 1;
    const ushort2* k2 = reinterpret_cast<const ushort2*>(lowkeys + w.start);
    const double2* v2 = reinterpret_cast<const double2*>(vals + w.start);
    for (int64_t i = threadIdx.x; i < npair; i += blockDim.x) {
      const ushort2 kk = k2[i];
      const double2 vv = v2[i];
      atomicAdd(&ltouch[kk.x], 1u);
      atomicAdd(&ltouch[kk.y], 1u);
      if (vv.x == vv.x) unsafeAtomicAdd(&lsums[kk.x], vv.x);
      if (vv.y == vv.y) unsafeAtomicAdd(&lsums[kk.y], vv.y);
    }
    if ((w.len & 1) && threadIdx.x == 0) {
      const int64_t i = end - 1;
      const int slot = lowkeys[i];
      atomicAdd(&ltouch[slot], 1u);
      const double v = vals[i];
      if (v == v) unsafeAtomicAdd(&lsums[slot], v);
    }
  }
  __syncthreads();
  const int64_t gbase = (int64_t)w.bucket << RL;
  for (int s = threadIdx.x; s < RANGE; s += blockDim.x) {
    const unsigned t = ltouch[s];
    if (!t || gbase + s >= n_slots) continue;
    unsafeAtomicAdd(&gsums[gbase + s], lsums[s]);
    atomicAdd(&growcnt[gbase + s], (unsigned long long)t);
  }
}

// ---------------- harness ----------------

static double run(const char* name, int reps, const std::function<void()>& pre,
                  const std::function<void()>& fn, double alg_bytes) {
  pre();
  fn();
  CHECK(hipDeviceSynchronize());
  hipEvent_t a, b;
  CHECK(hipEventCreate(&a));
  CHECK(hipEventCreate(&b));
  float total = 0;
  for (int r = 0; r < reps; ++r) {
    pre();
    CHECK(hipDeviceSynchronize());
    CHECK(hipEventRecord(a));
    fn();
    CHECK(hipEventRecord(b));
    CHECK(hipDeviceSynchronize());
    float ms;
    CHECK(hipEventElapsedTime(&ms, a, b));
    total += ms;
  }
  double per = total / reps;
  printf("%-34s %8.2f ms  alg16B %7.1f GB/s  %7.1f Grows/s\n", name, per,
         alg_bytes / per / 1e6, (alg_bytes / 16.0) / per / 1e6);
  fflush(stdout);
  return per;
}

int main(int argc, char** argv) {
  int64_t n = argc > 1 ? atoll(argv[1]) : 1000000000LL;
  const int64_t n_slots = 1000000;
  printf("n=%lld slots=%lld\n", (long long)n, (long long)n_slots);
  int64_t* keys;
  double* v0;
  CHECK(hipMalloc(&keys, n * 8));
  CHECK(hipMalloc(&v0, n * 8));
  {
    std::mt19937_64 gen(42);
    std::vector<int64_t> hk(1 << 24);
    std::vector<double> hv(1 << 24);
    for (int64_t off = 0; off < n; off += (int64_t)hk.size()) {
      int64_t m = std::min<int64_t>(hk.size(), n - off);
      for (int64_t i = 0; i < m; ++i) {
        hk[i] = gen() % n_slots;
        hv[i] = (double)(gen() % 1000) / 1000.0;
      }
      CHECK(hipMemcpy(keys + off, hk.data(), m * 8, hipMemcpyHostToDevice));
      CHECK(hipMemcpy(v0 + off, hv.data(), m * 8, hipMemcpyHostToDevice));
    }
  }
  double* r0;
  unsigned short* rk;
  unsigned* d_cur;
  double* gsums;
  unsigned long long* growcnt;
  const int64_t alloc_rows = (int64_t)(n * 1.25) + 4096 * 2048;
  CHECK(hipMalloc(&r0, alloc_rows * 8));
  CHECK(hipMalloc(&rk, alloc_rows * 2));
  CHECK(hipMalloc(&d_cur, 2048 * 4));
  CHECK(hipMalloc(&gsums, n_slots * 8));
  CHECK(hipMalloc(&growcnt, n_slots * 8));

  auto sweep = [&](auto rlTag) {
    constexpr int RL = decltype(rlTag)::value;
    const int nb = (int)((n_slots + (1 << RL) - 1) >> RL);
    // uniform keys: exact histogram close to n/nb; compute on host for regions
    std::vector<int64_t> h((size_t)nb, 0);
    {
      // approximate: uniform keys; region = n/nb * 1.1 margin, aligned
      // (probe only; production uses the exact histogram)
      int64_t per = (int64_t)((double)n / nb * 1.15) + 4096;
      per = (per + 63) & ~63LL;
      for (int b = 0; b < nb; ++b) h[b] = per;
    }
    std::vector<unsigned> cur((size_t)nb);
    std::vector<Work> work;
    int64_t off = 0;
    for (int b = 0; b < nb; ++b) {
      cur[b] = (unsigned)off;
      for (int64_t done = 0; done < h[b]; done += (1 << 21))
        work.push_back(Work{off + done, b,
                            (int32_t)std::min<int64_t>(1 << 21, h[b] - done)});
      off += h[b];
    }
    if (off > alloc_rows) { printf("alloc overflow\n"); exit(1); }
    Work* d_work;
    CHECK(hipMalloc(&d_work, work.size() * sizeof(Work)));
    CHECK(hipMemcpy(d_work, work.data(), work.size() * sizeof(Work),
                    hipMemcpyHostToDevice));
    auto reset_cur = [&] {
      CHECK(hipMemcpyAsync(d_cur, cur.data(), nb * 4, hipMemcpyHostToDevice, 0));
    };
    char nm[128];
    auto scat_reg = [&](auto rptTag) {
      constexpr int RPT = decltype(rptTag)::value;
      snprintf(nm, sizeof nm, "scat_reg RPT=%d RL=%d nb=%d", RPT, RL, nb);
      const int64_t ntiles = (n + BLOCK * RPT - 1) / (BLOCK * RPT);
      const uint32_t grid = (uint32_t)std::min<int64_t>(ntiles, 2048);
      run(nm, 2, reset_cur, [&] {
        hipLaunchKernelGGL((k_scat_reg<RPT, RL>), dim3(grid), dim3(BLOCK),
                           nb * 8, 0, keys, v0, n, n_slots, nb, d_cur, r0, rk);
      }, 26.0 * n);
    };
    scat_reg(std::integral_constant<int, 8>{});
    auto scat_lds = [&](auto rptTag, bool pscan) {
      constexpr int RPT = decltype(rptTag)::value;
      snprintf(nm, sizeof nm, "scat_lds%s RPT=%d RL=%d nb=%d",
               pscan ? "2" : "", RPT, RL, nb);
      const int64_t ntiles = (n + BLOCK * RPT - 1) / (BLOCK * RPT);
      const uint32_t grid = (uint32_t)std::min<int64_t>(ntiles, 2048);
      const uint32_t lds = BLOCK * RPT * 12 + nb * 16;
      run(nm, 2, reset_cur, [&] {
        if (pscan)
          hipLaunchKernelGGL((k_scat_lds2<RPT, RL>), dim3(grid), dim3(BLOCK),
                             lds, 0, keys, v0, n, n_slots, nb, d_cur, r0, rk);
        else
          hipLaunchKernelGGL((k_scat_lds<RPT, RL>), dim3(grid), dim3(BLOCK),
                             lds, 0, keys, v0, n, n_slots, nb, d_cur, r0, rk);
      }, 26.0 * n);
    };
    scat_lds(std::integral_constant<int, 16>{}, true);
    auto scat_lds3 = [&](auto rptTag, auto blkTag) {
      constexpr int RPT = decltype(rptTag)::value;
      constexpr int BLK = decltype(blkTag)::value;
      snprintf(nm, sizeof nm, "scat_lds3 RPT=%d BLK=%d RL=%d", RPT, BLK, RL);
      const int64_t tile_sz = (int64_t)BLK * RPT;
      const int64_t ntiles = (n + tile_sz - 1) / tile_sz;
      const uint32_t grid = (uint32_t)std::min<int64_t>(ntiles, 2048);
      const uint32_t lds = tile_sz * 12 + nb * 16 + 16;
      run(nm, 2, reset_cur, [&] {
        hipLaunchKernelGGL((k_scat_lds3<RPT, RL, BLK>), dim3(grid), dim3(BLK),
                           lds, 0, keys, v0, n, n_slots, nb, d_cur, r0, rk);
      }, 26.0 * n);
    };
    scat_lds3(std::integral_constant<int, 24>{}, std::integral_constant<int, 256>{});
    auto scat_lds4 = [&](auto rptTag, auto blkTag) {
      constexpr int RPT = decltype(rptTag)::value;
      constexpr int BLK = decltype(blkTag)::value;
      snprintf(nm, sizeof nm, "scat_lds4 RPT=%d BLK=%d RL=%d", RPT, BLK, RL);
      const int64_t tile_sz = (int64_t)BLK * RPT;
      const int64_t ntiles = (n + tile_sz - 1) / tile_sz;
      const uint32_t grid = (uint32_t)std::min<int64_t>(ntiles, 2048);
      const uint32_t lds = tile_sz * 12 + nb * 16 + 16;
      run(nm, 2, reset_cur, [&] {
        hipLaunchKernelGGL((k_scat_lds4<RPT, RL, BLK>), dim3(grid), dim3(BLK),
                           lds, 0, keys, v0, n, n_slots, nb, d_cur, r0, rk);
      }, 26.0 * n);
    };
    auto scat_lds5 = [&](auto rptTag, auto blkTag) {
      constexpr int RPT = decltype(rptTag)::value;
      constexpr int BLK = decltype(blkTag)::value;
      snprintf(nm, sizeof nm, "scat_lds5 RPT=%d BLK=%d RL=%d", RPT, BLK, RL);
      const int64_t tile_sz = (int64_t)BLK * RPT;
      const int64_t ntiles = (n + tile_sz - 1) / tile_sz;
      const uint32_t grid = (uint32_t)std::min<int64_t>(ntiles, 2048);
      const uint32_t lds = tile_sz * 12 + nb * 16 + 16;
      run(nm, 2, reset_cur, [&] {
        hipLaunchKernelGGL((k_scat_lds5<RPT, RL, BLK>), dim3(grid), dim3(BLK),
                           lds, 0, keys, v0, n, n_slots, nb, d_cur, r0, rk);
      }, 26.0 * n);
    };
    auto abl = [&](auto phTag) {
      constexpr int PH = decltype(phTag)::value;
      constexpr int RPT = 24, BLK = 256;
      snprintf(nm, sizeof nm, "scat_abl PH=%d RL=%d", PH, RL);
      const int64_t tile_sz = (int64_t)BLK * RPT;
      const int64_t ntiles = (n + tile_sz - 1) / tile_sz;
      const uint32_t grid = (uint32_t)std::min<int64_t>(ntiles, 2048);
      const uint32_t lds = tile_sz * 12 + nb * 16 + 16;
      run(nm, 2, reset_cur, [&] {
        hipLaunchKernelGGL((k_scat_abl<RPT, RL, BLK, PH>), dim3(grid),
                           dim3(BLK), lds, 0, keys, v0, n, n_slots, nb, d_cur,
                           r0, rk);
      }, 26.0 * n);
    };
    auto scat7 = [&](auto rptTag, auto blkTag) {
      constexpr int RPT = decltype(rptTag)::value;
      constexpr int BLK = decltype(blkTag)::value;
      constexpr int WPB = BLK / 64;
      snprintf(nm, sizeof nm, "scat_lds7 RPT=%d BLK=%d RL=%d", RPT, BLK, RL);
      const int64_t tile_sz = (int64_t)BLK * RPT;
      const int64_t ntiles = (n + tile_sz - 1) / tile_sz;
      const uint32_t grid = (uint32_t)std::min<int64_t>(ntiles, 2048);
      const uint32_t lds = tile_sz * 12 + nb * (2 * WPB + 2) * 4 + 16;
      run(nm, 2, reset_cur, [&] {
        hipLaunchKernelGGL((k_scat_lds7<RPT, RL, BLK>), dim3(grid), dim3(BLK),
                           lds, 0, keys, v0, n, n_slots, nb, d_cur, r0, rk);
      }, 26.0 * n);
    };
    scat7(std::integral_constant<int, 24>{}, std::integral_constant<int, 256>{});
    scat7(std::integral_constant<int, 16>{}, std::integral_constant<int, 512>{});
    scat_lds3(std::integral_constant<int, 24>{}, std::integral_constant<int, 512>{});
    auto scatw = [&](auto rptTag, auto blkTag) {
      constexpr int RPT = decltype(rptTag)::value;
      constexpr int BLK = decltype(blkTag)::value;
      constexpr int WPB = BLK / 64;
      snprintf(nm, sizeof nm, "scat_wave RPT=%d BLK=%d RL=%d", RPT, BLK, RL);
      const int64_t tile_sz = 64 * RPT;
      const int64_t ntiles = (n + tile_sz - 1) / tile_sz;
      const uint32_t grid =
          (uint32_t)std::min<int64_t>((ntiles + WPB - 1) / WPB, 4096);
      const uint32_t lds = WPB * (tile_sz * 12 + 3 * nb * 4 + 16);
      if (lds > 160 * 1024) { printf("%s: LDS too big (%u)\n", nm, lds); }
      else
        run(nm, 2, reset_cur, [&] {
          hipLaunchKernelGGL((k_scat_wave<RPT, RL, BLK>), dim3(grid), dim3(BLK),
                             lds, 0, keys, v0, n, n_slots, nb, d_cur, r0, rk);
        }, 26.0 * n);
    };
    scatw(std::integral_constant<int, 32>{}, std::integral_constant<int, 256>{});
    scatw(std::integral_constant<int, 32>{}, std::integral_constant<int, 384>{});
    scatw(std::integral_constant<int, 48>{}, std::integral_constant<int, 256>{});
    scatw(std::integral_constant<int, 16>{}, std::integral_constant<int, 512>{});
    abl(std::integral_constant<int, 31>{});   // full
    abl(std::integral_constant<int, 15>{});   // no writeout
    abl(std::integral_constant<int, 27>{});   // no reserve
    abl(std::integral_constant<int, 30>{});   // no ranks
    abl(std::integral_constant<int, 23>{});   // no stage
    abl(std::integral_constant<int, 29>{});   // no scan
    abl(std::integral_constant<int, 3>{});    // loads+ranks+scan only
    abl(std::integral_constant<int, 0>{});    // loads only
    // aggregate variants (consume whatever the last scatter left; perf-only)
    auto agg = [&](auto vecTag, auto blkTag) {
      constexpr int VEC = decltype(vecTag)::value;
      constexpr int BLK = decltype(blkTag)::value;
      snprintf(nm, sizeof nm, "agg VEC=%d BLK=%d RL=%d (%zu wi)", VEC, BLK, RL,
               work.size());
      auto nop = [] {};
      run(nm, 3, nop, [&] {
        hipLaunchKernelGGL((k_agg<RL, VEC, BLK>), dim3((uint32_t)work.size()),
                           dim3(BLK), 0, 0, r0, rk, d_work, n_slots, gsums,
                           growcnt);
      }, 10.0 * n);
    };
    // chunked pipeline: scatter+agg per ~CHUNK_ROWS slice, same scratch —
    // tests Infinity-Cache absorption of the bucket payload (L3-resident
    // between P1 and P2; HBM should only see the input reads)
    auto chunked = [&](int64_t chunk_rows) {
      snprintf(nm, sizeof nm, "chunked %lldM RL=%d", (long long)(chunk_rows / 1000000), RL);
      const int64_t cap_per_bucket =
          ((int64_t)(chunk_rows * 1.15 / nb) + 2048 + 63) & ~63LL;
      // work granularity sized so each chunk's agg fills the chip (~512+
      // blocks): items of cap/ceil(cap*nb/что... just target >= 512 items
      int64_t wchunk = (cap_per_bucket * nb + 511) / 512;
      wchunk = std::max<int64_t>(64, std::min<int64_t>(wchunk, 1 << 21)) & ~1LL;
      std::vector<unsigned> ccur((size_t)nb);
      std::vector<Work> cwork;
      for (int b = 0; b < nb; ++b) {
        ccur[b] = (unsigned)(b * cap_per_bucket);
        for (int64_t done = 0; done < cap_per_bucket; done += wchunk)
          cwork.push_back(Work{(int64_t)b * cap_per_bucket + done, b,
                               (int32_t)std::min<int64_t>(wchunk,
                                                          cap_per_bucket - done)});
      }
      if ((int64_t)nb * cap_per_bucket > alloc_rows) { printf("%s skip (alloc)\n", nm); return; }
      Work* d_cw;
      unsigned* d_ccur_init;
      CHECK(hipMalloc(&d_cw, cwork.size() * sizeof(Work)));
      CHECK(hipMemcpy(d_cw, cwork.data(), cwork.size() * sizeof(Work),
                      hipMemcpyHostToDevice));
      CHECK(hipMalloc(&d_ccur_init, nb * 4));
      CHECK(hipMemcpy(d_ccur_init, ccur.data(), nb * 4, hipMemcpyHostToDevice));
      // slack regions beyond each chunk's fill are read by the agg: make
      // sure they hold in-range lowkeys (first touch would be garbage)
      CHECK(hipMemset(rk, 0, alloc_rows * 2));
      auto nop = [] {};
      run(nm, 2, nop, [&] {
        for (int64_t c0 = 0; c0 < n; c0 += chunk_rows) {
          const int64_t cn = std::min(chunk_rows, n - c0);
          CHECK(hipMemcpyAsync(d_cur, d_ccur_init, nb * 4,
                               hipMemcpyDeviceToDevice, 0));
          const int64_t tile_sz = 256 * 24;
          const uint32_t grid = (uint32_t)std::min<int64_t>(
              (cn + tile_sz - 1) / tile_sz, 2048);
          const uint32_t lds = tile_sz * 12 + nb * 16 + 16;
          hipLaunchKernelGGL((k_scat_lds3<24, RL, 256>), dim3(grid), dim3(256),
                             lds, 0, keys + c0, v0 + c0, cn, n_slots, nb,
                             d_cur, r0, rk);
          hipLaunchKernelGGL((k_agg_u8<RL, 512>), dim3((uint32_t)cwork.size()),
                             dim3(512), 0, 0, r0, rk, d_cw, n_slots, gsums,
                             growcnt);
        }
      }, 26.0 * n);
      CHECK(hipFree(d_cw));
      CHECK(hipFree(d_ccur_init));
    };
    chunked(20000000);
    chunked(10000000);
    chunked(50000000);
    chunked(100000000);
    auto agg8 = [&](auto blkTag) {
      constexpr int BLK = decltype(blkTag)::value;
      snprintf(nm, sizeof nm, "agg_u8 BLK=%d RL=%d (%zu wi)", BLK, RL,
               work.size());
      auto nop = [] {};
      run(nm, 3, nop, [&] {
        hipLaunchKernelGGL((k_agg_u8<RL, BLK>), dim3((uint32_t)work.size()),
                           dim3(BLK), 0, 0, r0, rk, d_work, n_slots, gsums,
                           growcnt);
      }, 10.0 * n);
    };
    agg8(std::integral_constant<int, 512>{});
    agg8(std::integral_constant<int, 256>{});
    agg(std::integral_constant<int, 2>{}, std::integral_constant<int, 512>{});
    CHECK(hipFree(d_work));
  };
  sweep(std::integral_constant<int, 13>{});
  sweep(std::integral_constant<int, 12>{});
  printf("done\n");
  return 0;
}
