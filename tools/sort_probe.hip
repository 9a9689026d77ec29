// sort_probe — VERDICT r01 item 8: the sort scatter is the second-hottest
// kernel family (3 stable LSD passes, ~85-90 ms at 1e9 rows).  The r1
// full-LDS digit-major variant (8 waves, 140 KB) was 5x WORSE (occupancy).
// This probe isolates ONE pass and asks:
//   s1 (diagnostic): is the pass WRITE-bound?  Same compute, coalesced
//       (wrong-position) writes — the upper bound if scatter writes were
//       free.
//   s2: per-WAVE digit-major LDS staging (16 KB/wave slice; block of 4
//       waves = ~68 KB -> 2 blocks/CU): rows ranked by the same leader
//       loop but placed digit-major in LDS, then written out in ~64 B
//       digit runs instead of 8 B scatter.
//   s3: s2 but with 2 waves per block (36 KB -> 4 blocks/CU).
// Checked: s2/s3 output == s0 output bit-exact.
// Run: ./sort_probe [rows]
// Build: hipcc --offload-arch=gfx950 -O3 tools/sort_probe.hip -o tools/sort_probe
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <vector>
#include <algorithm>

#define CHECK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__); exit(1); } } while (0)

constexpr int BLOCK = 256;
constexpr int SORT_RPT = 32;
constexpr int SORT_TILE = 64 * SORT_RPT;
constexpr int SORT_WPB = 4;

__device__ __forceinline__ uint64_t mix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

__global__ void k_gen(unsigned long long* __restrict__ pairs, int64_t n,
                      uint64_t seed) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride)
    pairs[i] = ((unsigned long long)(unsigned)i << 32) |
               (unsigned)(mix64(i + seed) & 0xFFFFFu);
}

__global__ void __launch_bounds__(BLOCK) k_count(
    const unsigned long long* __restrict__ pairs, int64_t n, int shift,
    unsigned* __restrict__ C, int64_t ntiles) {
  __shared__ unsigned hist[SORT_WPB][256];
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  for (int64_t tile = (int64_t)blockIdx.x * SORT_WPB + wave; tile < ntiles;
       tile += (int64_t)gridDim.x * SORT_WPB) {
    for (int d = lane; d < 256; d += 64) hist[wave][d] = 0;
    __builtin_amdgcn_wave_barrier();
    const int64_t t0 = tile * SORT_TILE;
#pragma unroll 4
    for (int j = 0; j < SORT_RPT; ++j) {
      const int64_t row = t0 + (int64_t)j * 64 + lane;
      if (row < n)
        atomicAdd(&hist[wave][((unsigned)pairs[row] >> shift) & 255u], 1u);
    }
    __builtin_amdgcn_wave_barrier();
    for (int d = lane; d < 256; d += 64) C[tile * 256 + d] = hist[wave][d];
  }
}

__global__ void __launch_bounds__(1024) k_transpose256(
    const unsigned* __restrict__ C, unsigned* __restrict__ CT,
    int64_t ntiles) {
  __shared__ unsigned t[32][33];
  const int64_t tx0 = (int64_t)blockIdx.x * 32;
  const int dy = (int)(blockIdx.y * 32);
  const int lx = threadIdx.x & 31, ly = threadIdx.x >> 5;
  const int64_t src_row = tx0 + ly;
  if (src_row < ntiles) t[ly][lx] = C[src_row * 256 + dy + lx];
  __syncthreads();
  const int64_t dst_col = tx0 + lx;
  if (dst_col < ntiles) CT[(int64_t)(dy + ly) * ntiles + dst_col] = t[lx][ly];
}

// single-block exclusive scan over CT (256*ntiles u32 -> u64 offs)
__global__ void __launch_bounds__(1024) k_scan(
    const unsigned* __restrict__ CT, unsigned long long* __restrict__ offs,
    int64_t m) {
  __shared__ unsigned long long carry;
  __shared__ unsigned long long wsum[16];
  if (threadIdx.x == 0) carry = 0;
  __syncthreads();
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  for (int64_t base = 0; base < m; base += 1024) {
    const int64_t i = base + threadIdx.x;
    unsigned long long v = i < m ? CT[i] : 0;
    unsigned long long incl = v;
    for (int d = 1; d < 64; d <<= 1) {
      unsigned long long up = __shfl_up((long long)incl, d);
      if (lane >= d) incl += up;
    }
    if (lane == 63) wsum[wave] = incl;
    __syncthreads();
    if (wave == 0 && lane < 16) {
      unsigned long long w = wsum[lane];
      unsigned long long winc = w;
      for (int d = 1; d < 16; d <<= 1) {
        unsigned long long up = __shfl_up((long long)winc, d);
        if (lane >= d) winc += up;
      }
      wsum[lane] = winc - w;
    }
    __syncthreads();
    if (i < m) offs[i] = carry + wsum[wave] + incl - v;
    __syncthreads();
    if (threadIdx.x == 1023) carry += wsum[15] + incl;
    __syncthreads();
  }
}

// s0: production shape
template <bool COALESCED_DIAG>
__global__ void __launch_bounds__(BLOCK) k_scat0(
    const unsigned long long* __restrict__ pairs, int64_t n, int shift,
    const unsigned long long* __restrict__ offs, int64_t ntiles,
    unsigned long long* __restrict__ out) {
  __shared__ unsigned long long base[SORT_WPB][256];
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  for (int64_t tile = (int64_t)blockIdx.x * SORT_WPB + wave; tile < ntiles;
       tile += (int64_t)gridDim.x * SORT_WPB) {
    for (int d = lane; d < 256; d += 64)
      base[wave][d] = offs[(int64_t)d * ntiles + tile];
    __builtin_amdgcn_wave_barrier();
    const int64_t t0 = tile * SORT_TILE;
    for (int j = 0; j < SORT_RPT; ++j) {
      const int64_t row = t0 + (int64_t)j * 64 + lane;
      const bool valid = row < n;
      const unsigned long long p = valid ? pairs[row] : 0;
      const unsigned d = ((unsigned)p >> shift) & 255u;
      unsigned long long exec = __ballot(valid);
      unsigned long long pos = 0;
      while (exec) {
        const int leader = __ffsll((long long)exec) - 1;
        const unsigned dl = (unsigned)__shfl((int)d, leader);
        const unsigned long long members = __ballot(valid && d == dl);
        if (valid && d == dl) {
          const unsigned rank =
              (unsigned)__popcll(members & ((1ULL << lane) - 1ULL));
          pos = base[wave][dl] + rank;
        }
        if (lane == leader) base[wave][dl] += __popcll(members);
        exec &= ~members;
      }
      if (valid) {
        if (COALESCED_DIAG)
          out[row] = p + pos;  // wrong result, same compute: write bound?
        else
          out[pos] = p;
      }
      __builtin_amdgcn_wave_barrier();
    }
  }
}

// s2/s3: per-wave digit-major LDS staging (WPB waves per block, each
// wave owns a 2048-row tile and a 16 KB slice), then coalesced ~64 B
// digit-run writeout — the leader-loop ranking is unchanged, only the
// write pattern differs
template <int WPB>
__global__ void __launch_bounds__(64 * WPB) k_scat_staged2(
    const unsigned long long* __restrict__ pairs, int64_t n, int shift,
    const unsigned long long* __restrict__ offs,
    const unsigned* __restrict__ C, int64_t ntiles,
    unsigned long long* __restrict__ out) {
  __shared__ unsigned long long stage[WPB][SORT_TILE];
  __shared__ unsigned dstart[WPB][256];
  __shared__ unsigned long long gbase[WPB][256];
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  for (int64_t tile = (int64_t)blockIdx.x * WPB + wave; tile < ntiles;
       tile += (int64_t)gridDim.x * WPB) {
    // tile-local exclusive digit starts from C[tile][256]
    unsigned carry = 0;
    for (int q = 0; q < 4; ++q) {
      const int d = q * 64 + lane;
      unsigned v = C[tile * 256 + d];
      unsigned incl = v;
      for (int s = 1; s < 64; s <<= 1) {
        unsigned up = __shfl_up(incl, s);
        if (lane >= s) incl += up;
      }
      dstart[wave][d] = carry + incl - v;
      gbase[wave][d] = offs[(int64_t)d * ntiles + tile];
      carry += __shfl(incl, 63);
    }
    __builtin_amdgcn_wave_barrier();
    // rank + stage digit-major into this wave's LDS slice
    unsigned fill[4];
    for (int q = 0; q < 4; ++q) fill[q] = 0;  // per-digit fill counters in
    // registers won't work (digits arbitrary); use LDS cursors instead:
    // reuse dstart as the running cursor (restore not needed afterwards)
    const int64_t t0 = tile * SORT_TILE;
    (void)fill;
    for (int j = 0; j < SORT_RPT; ++j) {
      const int64_t row = t0 + (int64_t)j * 64 + lane;
      const bool valid = row < n;
      const unsigned long long p = valid ? pairs[row] : 0;
      const unsigned d = ((unsigned)p >> shift) & 255u;
      unsigned long long exec = __ballot(valid);
      unsigned pos = 0;
      while (exec) {
        const int leader = __ffsll((long long)exec) - 1;
        const unsigned dl = (unsigned)__shfl((int)d, leader);
        const unsigned long long members = __ballot(valid && d == dl);
        if (valid && d == dl) {
          const unsigned rank =
              (unsigned)__popcll(members & ((1ULL << lane) - 1ULL));
          pos = dstart[wave][dl] + rank;
        }
        if (lane == leader) dstart[wave][dl] += __popcll(members);
        exec &= ~members;
      }
      if (valid) stage[wave][pos] = p;
      __builtin_amdgcn_wave_barrier();
    }
    // writeout: LDS is digit-major; consecutive lanes hit consecutive
    // positions of (mostly) the same digit — ~64 B runs per digit
    const int tlen = (int)min((int64_t)SORT_TILE, n - t0);
    for (int p2 = lane; p2 < tlen; p2 += 64) {
      const unsigned long long v = stage[wave][p2];
      const unsigned d = ((unsigned)v >> shift) & 255u;
      // position within digit run: p2 - (digit start BEFORE cursors moved)
      // dstart has been advanced to the digit END; start = end - count
      const unsigned cend = dstart[wave][d];
      const unsigned ccount = C[tile * 256 + d];
      out[gbase[wave][d] + (p2 - (cend - ccount))] = v;
    }
    __builtin_amdgcn_wave_barrier();
  }
}

// s4: bit-ballot ranking — the serial leader loop (one iteration per
// DISTINCT digit in the wave, ~50 for 256 digits) becomes 8 ballots:
// lanes sharing this lane's digit = AND over digit bits of
// (bit set ? ballot : ~ballot).  Rank = popcount(below); the lowest lane
// of each group advances the digit base.  O(8) per row, no divergent loop.
__global__ void __launch_bounds__(BLOCK) k_scat_ballot(
    const unsigned long long* __restrict__ pairs, int64_t n, int shift,
    const unsigned long long* __restrict__ offs, int64_t ntiles,
    unsigned long long* __restrict__ out) {
  __shared__ unsigned long long base[SORT_WPB][256];
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const unsigned long long below = (1ull << lane) - 1ull;
  for (int64_t tile = (int64_t)blockIdx.x * SORT_WPB + wave; tile < ntiles;
       tile += (int64_t)gridDim.x * SORT_WPB) {
    for (int d = lane; d < 256; d += 64)
      base[wave][d] = offs[(int64_t)d * ntiles + tile];
    __builtin_amdgcn_wave_barrier();
    const int64_t t0 = tile * SORT_TILE;
    for (int j = 0; j < SORT_RPT; ++j) {
      const int64_t row = t0 + (int64_t)j * 64 + lane;
      const bool valid = row < n;
      const unsigned long long p = valid ? pairs[row] : 0;
      const unsigned d = ((unsigned)p >> shift) & 255u;
      unsigned long long same = __ballot(valid);
#pragma unroll
      for (int b = 0; b < 8; ++b) {
        const unsigned long long m = __ballot((d >> b) & 1u);
        same &= ((d >> b) & 1u) ? m : ~m;
      }
      if (valid) {
        const unsigned rank = (unsigned)__popcll(same & below);
        const unsigned long long pos = base[wave][d] + rank;
        // lowest member advances the base for the next step
        if (lane == __ffsll((long long)same) - 1)
          base[wave][d] += __popcll(same);
        out[pos] = p;
      }
      __builtin_amdgcn_wave_barrier();
    }
  }
}

__global__ void k_cmp64(const unsigned long long* a,
                        const unsigned long long* b, int64_t n,
                        unsigned long long* bad) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride)
    if (a[i] != b[i]) atomicAdd(bad, 1ULL);
}

int main(int argc, char** argv) {
  const int64_t N = argc > 1 ? atoll(argv[1]) : 1000000000LL;
  const int64_t ntiles = (N + SORT_TILE - 1) / SORT_TILE;
  printf("sort_probe: N=%lld ntiles=%lld\n", (long long)N, (long long)ntiles);
  unsigned long long *pairs, *out, *ref;
  unsigned *C, *CT;
  unsigned long long *offs, *d_bad;
  CHECK(hipMalloc(&pairs, N * 8));
  CHECK(hipMalloc(&out, N * 8));
  CHECK(hipMalloc(&ref, N * 8));
  CHECK(hipMalloc(&C, ntiles * 256 * 4));
  CHECK(hipMalloc(&CT, ntiles * 256 * 4));
  CHECK(hipMalloc(&offs, ntiles * 256 * 8));
  CHECK(hipMalloc(&d_bad, 8));
  hipLaunchKernelGGL(k_gen, dim3(4096), dim3(BLOCK), 0, 0, pairs, N, 7);
  const int shift = 8;
  hipLaunchKernelGGL(k_count, dim3(2048), dim3(BLOCK), 0, 0, pairs, N,
                     shift, C, ntiles);
  hipLaunchKernelGGL(k_transpose256,
                     dim3((uint32_t)((ntiles + 31) / 32), 8), dim3(1024), 0,
                     0, C, CT, ntiles);
  hipLaunchKernelGGL(k_scan, dim3(1), dim3(1024), 0, 0, CT, offs,
                     ntiles * 256);
  CHECK(hipDeviceSynchronize());

  hipEvent_t e0, e1;
  CHECK(hipEventCreate(&e0));
  CHECK(hipEventCreate(&e1));
  auto run = [&](const char* name, auto launch, bool check) {
    float best = 1e30f;
    for (int r = 0; r < 4; ++r) {
      CHECK(hipDeviceSynchronize());
      CHECK(hipEventRecord(e0, 0));
      launch();
      CHECK(hipEventRecord(e1, 0));
      CHECK(hipDeviceSynchronize());
      float ms;
      hipEventElapsedTime(&ms, e0, e1);
      if (r > 0) best = std::min(best, ms);
    }
    const char* chk = "";
    if (check) {
      CHECK(hipMemset(d_bad, 0, 8));
      hipLaunchKernelGGL(k_cmp64, dim3(4096), dim3(BLOCK), 0, 0, out, ref,
                         N, d_bad);
      unsigned long long bad;
      CHECK(hipMemcpy(&bad, d_bad, 8, hipMemcpyDeviceToHost));
      if (bad) chk = "  !! MISMATCH";
    }
    printf("%-22s: %8.3f ms (%.2f TB/s on 16GB R+W)%s\n", name, best,
           16.0 / best, chk);
  };

  run("s0 current", [&] {
    hipLaunchKernelGGL(k_scat0<false>, dim3(2048), dim3(BLOCK), 0, 0, pairs,
                       N, shift, offs, ntiles, ref);
  }, false);
  run("s1 coalesced-diag", [&] {
    hipLaunchKernelGGL(k_scat0<true>, dim3(2048), dim3(BLOCK), 0, 0, pairs,
                       N, shift, offs, ntiles, out);
  }, false);
  run("s2 staged wpb4", [&] {
    hipLaunchKernelGGL(k_scat_staged2<4>, dim3(2048), dim3(256), 0, 0,
                       pairs, N, shift, offs, C, ntiles, out);
  }, true);
  run("s3 staged wpb2", [&] {
    hipLaunchKernelGGL(k_scat_staged2<2>, dim3(4096), dim3(128), 0, 0,
                       pairs, N, shift, offs, C, ntiles, out);
  }, true);
  run("s4 bit-ballot", [&] {
    hipLaunchKernelGGL(k_scat_ballot, dim3(2048), dim3(BLOCK), 0, 0, pairs,
                       N, shift, offs, ntiles, out);
  }, true);
  run("s4 g4096", [&] {
    hipLaunchKernelGGL(k_scat_ballot, dim3(4096), dim3(BLOCK), 0, 0, pairs,
                       N, shift, offs, ntiles, out);
  }, true);
  printf("done\n");
  return 0;
}
