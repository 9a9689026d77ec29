// hipframe.hip — gfx950 (MI355X, CDNA4) implementation of the hipframe C-ABI.
//
// Every op here is the device form of one Modin operator template (the
// reference executes these as pandas calls inside each partition — see the
// per-function citations in include/hipframe.h).  All paths are
// HBM-bandwidth-bound (no MFMA anywhere on this path), so the
// kernels are built around the CDNA4 streaming rules:
//   - 16 B/lane vectorized loads (double2 / longlong2) — the coalescing sweet
//     spot on gfx950,
//   - grid-stride loops capped at a few thousand 256-thread workgroups
//     (≫256 so all 8 XCDs fill),
//   - wave64 shuffle reductions + LDS block reductions, one atomic per block,
//   - dense-key groupby via hardware global_atomic_add_f64 into an
//     HBM/L3-resident key-indexed table (the table for the north-star config,
//     1e6 keys, is 8–24 MB: Infinity-Cache resident).
//
// Build: hipcc --offload-arch=gfx950 -O3 -shared -fPIC (see Makefile).

#include <hip/hip_runtime.h>
#include <algorithm>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <cstdlib>
#include <string>
#include <type_traits>
#include <unordered_map>
#include <vector>

#include "../../include/hipframe.h"

// ---------------------------------------------------------------------------
// module state & error plumbing
// ---------------------------------------------------------------------------

struct hf_col {
  void*   dptr;
  int64_t len;
  int     dtype;
  int     gpu;
  // cached per-bucket histogram for the radix groupby path (device u64
  // array of n_buckets entries).  Valid when hist_kmin/hist_nb match the
  // current query.  The column is immutable, so this is the device analog
  // of the reference's lazy-metadata caches (modin/core/dataframe/pandas/
  // metadata/) — a repeated groupby on the same key column skips the
  // histogram pass.
  void*   d_hist = nullptr;
  int64_t hist_kmin = 0;
  int64_t hist_nb = 0;
  // cached u32 shifted copy of an int64 key column (key - key_min), the
  // narrow-key fast path for the radix scatter (12 B/row instead of 16).
  // Same immutable-column caching rationale as d_hist.
  void*   d_k32 = nullptr;
  int64_t k32_min = 0;
  // cached radix-scatter layout derived from d_hist: device cursor-init
  // array (region starts) + device work items — steady-state groupby
  // launches copy cursors D2D and never touch the host (no H2D, no sync)
  void*   d_curinit = nullptr;   // u32[nb]
  void*   d_work = nullptr;      // GbWorkItem[work_n]
  int64_t work_n = 0;
  int64_t radix_rows = 0;        // payload rows (64-aligned region total)
  int     radix_rl = 0;          // validity: matches (hist_kmin, hist_nb)
};

namespace {

thread_local std::string g_err;

struct TimedPair { hipEvent_t a, b; std::string name; };

struct State {
  bool        inited = false;
  int         gpu = -1;
  hipStream_t stream = nullptr;
  // consumer stream for the overlapped radix-groupby P2 (all OTHER compute
  // stays on `stream`; stream2 work is always fenced back onto `stream`
  // with an event before any buffer it read is freed, so the allocator's
  // single-stream reuse argument still holds)
  hipStream_t stream2 = nullptr;
  hipEvent_t  ov_ev[10] = {};
  // small persistent device scratch: reduce accumulators + error word +
  // compact bookkeeping
  void*       d_scratch = nullptr;   // see layout below
  // profiling
  bool        profiling = false;
  std::vector<TimedPair> pending;
  std::unordered_map<std::string, std::pair<int64_t, double>> stats;
};
State g;

// d_scratch layout (bytes):
//   [0..128)    reduce accumulator block (hf_reduce)
//   [128..136)  groupby error word (u64: count of out-of-range keys)
//   [136..144)  compact total (i64 n_groups)
constexpr int64_t SCRATCH_BYTES = 4096;
constexpr int64_t SCRATCH_GB_ERR = 128;
constexpr int64_t SCRATCH_NGROUPS = 136;
constexpr int64_t SCRATCH_JOIN_HIST_ERR = 144;
constexpr int64_t SCRATCH_JOIN_FIXUP_ERR = 152;
constexpr int64_t SCRATCH_HASH_FULL = 160;

int set_err(int code, const char* where, const char* what) {
  g_err = std::string(where) + ": " + what;
  return code;
}

int set_hip_err(const char* where, hipError_t e) {
  return set_err(HF_ERR_HIP, where, hipGetErrorString(e));
}

#define HF_HIP(where, call)                                   \
  do {                                                        \
    hipError_t _e = (call);                                   \
    if (_e != hipSuccess) return set_hip_err(where, _e);      \
  } while (0)

#define HF_NEED_INIT(where)                                   \
  if (!g.inited) return set_err(HF_ERR_NOINIT, where, "hf_init not called")

int64_t dtype_size(int dt) {
  switch (dt) {
    case HF_INT64:   return 8;
    case HF_FLOAT64: return 8;
    default:         return 0;
  }
}

// resolve pending profiling events into stats (syncs the stream)
int resolve_stats(const char* where) {
  if (g.pending.empty()) return HF_OK;
  HF_HIP(where, hipStreamSynchronize(g.stream));
  for (auto& p : g.pending) {
    float ms = 0.f;
    hipEventElapsedTime(&ms, p.a, p.b);
    auto& s = g.stats[p.name];
    s.first += 1;
    s.second += ms;
    hipEventDestroy(p.a);
    hipEventDestroy(p.b);
  }
  g.pending.clear();
  return HF_OK;
}

// launch helper with optional event bracketing; pass the stream the launch
// targets (default: the module stream)
template <typename F>
int timed_launch_on(const char* name, hipStream_t s, F&& launch) {
  if (!g.profiling) {
    launch();
    hipError_t e = hipGetLastError();
    if (e != hipSuccess) return set_hip_err(name, e);
    return HF_OK;
  }
  TimedPair p;
  p.name = name;
  HF_HIP(name, hipEventCreate(&p.a));
  HF_HIP(name, hipEventCreate(&p.b));
  HF_HIP(name, hipEventRecord(p.a, s));
  launch();
  hipError_t e = hipGetLastError();
  if (e != hipSuccess) return set_hip_err(name, e);
  HF_HIP(name, hipEventRecord(p.b, s));
  g.pending.push_back(p);
  if (g.pending.size() > 4096) return resolve_stats(name);
  return HF_OK;
}

template <typename F>
int timed_launch(const char* name, F&& launch) {
  return timed_launch_on(name, g.stream, std::forward<F>(launch));
}

constexpr int BLOCK = 256;
// memory-bound grid cap: ≫256 workgroups to fill 8 XCDs, grid-stride the rest
// (cdna_hip_programming.md §6 Guideline 11)
constexpr int64_t GRID_CAP = 4096;

int64_t grid_for(int64_t work_items) {
  int64_t b = (work_items + BLOCK - 1) / BLOCK;
  if (b < 1) b = 1;
  return b < GRID_CAP ? b : GRID_CAP;
}

}  // namespace


// Device allocator: a size-bucketed cache over hipMalloc.  All compute runs
// on ONE module stream, so reusing a cached block is ordered-correct by
// construction (every prior user's work precedes the next user's on the
// stream).  ROCm 7.2's hipMallocAsync default pool showed reuse corruption
// under the join's varied alloc/free pattern (partial histograms + "write
// access to a read-only page" faults, joinbench vs joindbg bisect) — this
// cache replaces it.  Steady-state workloads allocate nothing.
// HF_SYNC_ALLOC=1 bypasses the cache (plain hipMalloc/hipFree, debug).
namespace {
inline bool sync_alloc() {
  static int v = -1;
  if (v < 0) v = getenv("HF_SYNC_ALLOC") ? 1 : 0;
  return v == 1;
}

struct DevCache {
  std::unordered_map<int64_t, std::vector<void*>> free_by_size;
  std::unordered_map<void*, int64_t> size_of;
  int64_t cached_bytes = 0;

  hipError_t alloc(void** p, int64_t bytes) {
    bytes = (bytes + 255) & ~255LL;
    auto it = free_by_size.find(bytes);
    if (it != free_by_size.end() && !it->second.empty()) {
      *p = it->second.back();
      it->second.pop_back();
      cached_bytes -= bytes;
      return hipSuccess;
    }
    hipError_t e = hipMalloc(p, bytes);
    if (e == hipErrorOutOfMemory) {
      trim();
      e = hipMalloc(p, bytes);
    }
    if (e == hipSuccess) size_of[*p] = bytes;
    return e;
  }
  void free(void* p) {
    auto it = size_of.find(p);
    if (it == size_of.end()) { hipFree(p); return; }
    free_by_size[it->second].push_back(p);
    cached_bytes += it->second;
  }
  void trim() {  // OOM fallback: release every cached block after draining
                 // outstanding users
    hipDeviceSynchronize();
    for (auto& kv : free_by_size)
      for (void* p : kv.second) { size_of.erase(p); hipFree(p); }
    free_by_size.clear();
    cached_bytes = 0;
  }
};
DevCache g_cache;

inline hipError_t dev_alloc(void** p, int64_t bytes, hipStream_t s) {
  if (sync_alloc()) { hipStreamSynchronize(s); return hipMalloc(p, bytes); }
  return g_cache.alloc(p, bytes);
}
inline hipError_t dev_free(void* p, hipStream_t s) {
  if (sync_alloc()) { hipStreamSynchronize(s); return hipFree(p); }
  g_cache.free(p);
  return hipSuccess;
}
}  // namespace

// ---------------------------------------------------------------------------
// kernels: Map (elementwise scalar) — algebra/map.py:28 device form
// ---------------------------------------------------------------------------

namespace {

template <int OP>
__device__ __forceinline__ double map1_f64(double x, double s) {
  switch (OP) {
    case HF_MAP_ADD:    return x + s;
    case HF_MAP_SUB:    return x - s;
    case HF_MAP_RSUB:   return s - x;
    case HF_MAP_MUL:    return x * s;
    case HF_MAP_DIV:    return x / s;
    case HF_MAP_RDIV:   return s / x;
    case HF_MAP_FILLNA: return (x != x) ? s : x;
    case HF_MAP_ABS:    return fabs(x);
    case HF_MAP_NEG:    return -x;
    case HF_MAP_SQRT:   return sqrt(x);
    // NaN-PROPAGATING (pandas clip leaves NaN alone; fmin/fmax would
    // replace NaN with the bound)
    case HF_MAP_ROUND:  return rint(x * s) / s;
    case HF_MAP_MIN:    return (x != x) ? x : fmin(x, s);
    case HF_MAP_MAX:    return (x != x) ? x : fmax(x, s);
  }
  return x;
}

template <int OP>
__global__ void __launch_bounds__(BLOCK) k_map_f64(const double* __restrict__ in,
                                                   double* __restrict__ out,
                                                   double s, int64_t n) {
  // 16 B/lane double2 stream; tail element handled by thread 0 of block 0.
  const int64_t npair = n >> 1;
  const double2* in2 = reinterpret_cast<const double2*>(in);
  double2* out2 = reinterpret_cast<double2*>(out);
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < npair; i += stride) {
    double2 v = in2[i];
    v.x = map1_f64<OP>(v.x, s);
    v.y = map1_f64<OP>(v.y, s);
    out2[i] = v;
  }
  if ((n & 1) && blockIdx.x == 0 && threadIdx.x == 0)
    out[n - 1] = map1_f64<OP>(in[n - 1], s);
}

template <int OP>
__device__ __forceinline__ int64_t map1_i64(int64_t x, int64_t s) {
  switch (OP) {
    case HF_MAP_ADD:  return x + s;
    case HF_MAP_SUB:  return x - s;
    case HF_MAP_RSUB: return s - x;
    case HF_MAP_MUL:  return x * s;
    case HF_MAP_ABS:  return x < 0 ? -x : x;
    case HF_MAP_MIN:  return x < s ? x : s;
    case HF_MAP_MAX:  return x > s ? x : s;
    case HF_MAP_NEG:  return -x;
    case HF_MAP_IDIV: {  // Python floordiv: round toward -inf
      int64_t q = x / s, r = x % s;
      return (r != 0 && ((r < 0) != (s < 0))) ? q - 1 : q;
    }
    case HF_MAP_IMOD: {  // Python mod: result takes s's sign
      int64_t r = x % s;
      return (r != 0 && ((r < 0) != (s < 0))) ? r + s : r;
    }
  }
  return x;
}

template <int OP>
__global__ void __launch_bounds__(BLOCK) k_map_i64(const int64_t* __restrict__ in,
                                                   int64_t* __restrict__ out,
                                                   int64_t s, int64_t n) {
  const int64_t npair = n >> 1;
  const longlong2* in2 = reinterpret_cast<const longlong2*>(in);
  longlong2* out2 = reinterpret_cast<longlong2*>(out);
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < npair; i += stride) {
    longlong2 v = in2[i];
    v.x = map1_i64<OP>(v.x, s);
    v.y = map1_i64<OP>(v.y, s);
    out2[i] = v;
  }
  if ((n & 1) && blockIdx.x == 0 && threadIdx.x == 0)
    out[n - 1] = map1_i64<OP>(in[n - 1], s);
}

__global__ void __launch_bounds__(BLOCK) k_cast_i64_f64(const int64_t* __restrict__ in,
                                                        double* __restrict__ out,
                                                        int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) out[i] = (double)in[i];
}

__global__ void __launch_bounds__(BLOCK) k_cast_f64_i64(const double* __restrict__ in,
                                                        int64_t* __restrict__ out,
                                                        int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) out[i] = (int64_t)in[i];
}

// ---------------------------------------------------------------------------
// kernels: Binary (elementwise column op column) — algebra/binary.py device form
// ---------------------------------------------------------------------------

__device__ __forceinline__ double hf_minv(double a, double b) {
  return fmin(a, b);  // NaN-skipping: pandas axis=1 min/max fold
}
__device__ __forceinline__ int64_t hf_minv(int64_t a, int64_t b) {
  return a < b ? a : b;
}
__device__ __forceinline__ double hf_maxv(double a, double b) {
  return fmax(a, b);
}
__device__ __forceinline__ int64_t hf_maxv(int64_t a, int64_t b) {
  return a > b ? a : b;
}

template <int OP, typename T>
__device__ __forceinline__ T bin1(T a, T b) {
  switch (OP) {
    case HF_BIN_ADD: return a + b;
    case HF_BIN_SUB: return a - b;
    case HF_BIN_MUL: return a * b;
    case HF_BIN_DIV: return a / b;
    case HF_BIN_MIN: return hf_minv(a, b);
    case HF_BIN_MAX: return hf_maxv(a, b);
  }
  return a;
}

template <int OP, typename T, typename T2>
__global__ void __launch_bounds__(BLOCK) k_bin(const T* __restrict__ a,
                                               const T* __restrict__ b,
                                               T* __restrict__ out, int64_t n) {
  const int64_t npair = n >> 1;
  const T2* a2 = reinterpret_cast<const T2*>(a);
  const T2* b2 = reinterpret_cast<const T2*>(b);
  T2* o2 = reinterpret_cast<T2*>(out);
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < npair; i += stride) {
    T2 va = a2[i], vb = b2[i];
    va.x = bin1<OP, T>(va.x, vb.x);
    va.y = bin1<OP, T>(va.y, vb.y);
    o2[i] = va;
  }
  if ((n & 1) && blockIdx.x == 0 && threadIdx.x == 0)
    out[n - 1] = bin1<OP, T>(a[n - 1], b[n - 1]);
}

// ---------------------------------------------------------------------------
// kernels: TreeReduce — algebra/tree_reduce.py device form
// pandas nan-skipping sum/count/min/max in one pass.
// Wave64 shuffle reduce -> LDS across waves -> one CAS/atomic per block
// (cdna_hip_programming.md Appendix B "Reduction").
// ---------------------------------------------------------------------------

struct ReduceAccF64 {  // lives in d_scratch[0..128)
  double sum;
  unsigned long long count;
  double mn, mx;
};
struct ReduceAccI64 {
  long long sum;
  unsigned long long count;
  long long mn, mx;
};

__device__ void atomic_min_f64(double* addr, double v) {
  unsigned long long* p = reinterpret_cast<unsigned long long*>(addr);
  unsigned long long old = *p, assumed;
  while (v < __longlong_as_double(old)) {
    assumed = old;
    old = atomicCAS(p, assumed, __double_as_longlong(v));
    if (old == assumed) break;
  }
}
__device__ void atomic_max_f64(double* addr, double v) {
  unsigned long long* p = reinterpret_cast<unsigned long long*>(addr);
  unsigned long long old = *p, assumed;
  while (v > __longlong_as_double(old)) {
    assumed = old;
    old = atomicCAS(p, assumed, __double_as_longlong(v));
    if (old == assumed) break;
  }
}
__device__ void atomic_min_i64(long long* addr, long long v) {
  unsigned long long* p = reinterpret_cast<unsigned long long*>(addr);
  unsigned long long old = *p, assumed;
  while (v < (long long)old) {
    assumed = old;
    old = atomicCAS(p, assumed, (unsigned long long)v);
    if (old == assumed) break;
  }
}
__device__ void atomic_max_i64(long long* addr, long long v) {
  unsigned long long* p = reinterpret_cast<unsigned long long*>(addr);
  unsigned long long old = *p, assumed;
  while (v > (long long)old) {
    assumed = old;
    old = atomicCAS(p, assumed, (unsigned long long)v);
    if (old == assumed) break;
  }
}

__global__ void k_reduce_init(ReduceAccF64* f, ReduceAccI64* i64a) {
  f->sum = 0.0; f->count = 0;
  f->mn = __longlong_as_double(0x7FF0000000000000LL);   // +inf
  f->mx = __longlong_as_double(0xFFF0000000000000LL);   // -inf
  i64a->sum = 0; i64a->count = 0;
  i64a->mn = 0x7FFFFFFFFFFFFFFFLL;
  i64a->mx = 0x8000000000000000LL;
}

__global__ void __launch_bounds__(BLOCK) k_reduce_f64(const double* __restrict__ in,
                                                      int64_t n, ReduceAccF64* acc) {
  double sum = 0.0, mn = __longlong_as_double(0x7FF0000000000000LL),
         mx = __longlong_as_double(0xFFF0000000000000LL);
  unsigned long long cnt = 0;
  const int64_t npair = n >> 1;
  const double2* in2 = reinterpret_cast<const double2*>(in);
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < npair; i += stride) {
    double2 v = in2[i];
    if (v.x == v.x) { sum += v.x; ++cnt; mn = fmin(mn, v.x); mx = fmax(mx, v.x); }
    if (v.y == v.y) { sum += v.y; ++cnt; mn = fmin(mn, v.y); mx = fmax(mx, v.y); }
  }
  if ((n & 1) && blockIdx.x == 0 && threadIdx.x == 0) {
    double v = in[n - 1];
    if (v == v) { sum += v; ++cnt; mn = fmin(mn, v); mx = fmax(mx, v); }
  }
  // wave64 shuffle reduce
  for (int off = 32; off > 0; off >>= 1) {
    sum += __shfl_down(sum, off);
    cnt += __shfl_down(cnt, off);
    mn = fmin(mn, __shfl_down(mn, off));
    mx = fmax(mx, __shfl_down(mx, off));
  }
  __shared__ double s_sum[BLOCK / 64], s_mn[BLOCK / 64], s_mx[BLOCK / 64];
  __shared__ unsigned long long s_cnt[BLOCK / 64];
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) { s_sum[wave] = sum; s_cnt[wave] = cnt; s_mn[wave] = mn; s_mx[wave] = mx; }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < BLOCK / 64; ++w) {
      sum += s_sum[w]; cnt += s_cnt[w];
      mn = fmin(mn, s_mn[w]); mx = fmax(mx, s_mx[w]);
    }
    if (cnt) {
      unsafeAtomicAdd(&acc->sum, sum);
      atomicAdd(&acc->count, cnt);
      atomic_min_f64(&acc->mn, mn);
      atomic_max_f64(&acc->mx, mx);
    }
  }
}

__device__ __forceinline__ long long llmin(long long a, long long b) { return a < b ? a : b; }
__device__ __forceinline__ long long llmax(long long a, long long b) { return a > b ? a : b; }

__global__ void __launch_bounds__(BLOCK) k_reduce_i64(const int64_t* __restrict__ in,
                                                      int64_t n, ReduceAccI64* acc) {
  long long sum = 0, mn = 0x7FFFFFFFFFFFFFFFLL, mx = 0x8000000000000000LL;
  unsigned long long cnt = 0;
  const int64_t npair = n >> 1;
  const longlong2* in2 = reinterpret_cast<const longlong2*>(in);
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < npair; i += stride) {
    longlong2 v = in2[i];
    sum += v.x + v.y; cnt += 2;
    mn = llmin(mn, llmin(v.x, v.y));
    mx = llmax(mx, llmax(v.x, v.y));
  }
  if ((n & 1) && blockIdx.x == 0 && threadIdx.x == 0) {
    long long v = in[n - 1];
    sum += v; ++cnt; mn = llmin(mn, v); mx = llmax(mx, v);
  }
  for (int off = 32; off > 0; off >>= 1) {
    sum += __shfl_down(sum, off);
    cnt += __shfl_down(cnt, off);
    mn = llmin(mn, (long long)__shfl_down(mn, off));
    mx = llmax(mx, (long long)__shfl_down(mx, off));
  }
  __shared__ long long s_sum[BLOCK / 64], s_mn[BLOCK / 64], s_mx[BLOCK / 64];
  __shared__ unsigned long long s_cnt[BLOCK / 64];
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) { s_sum[wave] = sum; s_cnt[wave] = cnt; s_mn[wave] = mn; s_mx[wave] = mx; }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < BLOCK / 64; ++w) {
      sum += s_sum[w]; cnt += s_cnt[w];
      mn = llmin(mn, s_mn[w]); mx = llmax(mx, s_mx[w]);
    }
    if (cnt) {
      atomicAdd((unsigned long long*)&acc->sum, (unsigned long long)sum);
      atomicAdd(&acc->count, cnt);
      atomic_min_i64(&acc->mn, mn);
      atomic_max_i64(&acc->mx, mx);
    }
  }
}

// ---------------------------------------------------------------------------
// kernels: GroupByReduce — algebra/groupby.py:124/:211 device form.
// Dense key-indexed table accumulation: keys in [key_min, key_min+n_slots).
// One u32-slot rowcnt atomic per row + one hardware f64 atomic add per
// non-NaN value.  The 1e6-key north-star table (8 MB sums + 8 MB rowcnt)
// is Infinity-Cache resident; atomics execute memory-side so the per-XCD L2
// incoherence is not in play.
// ---------------------------------------------------------------------------

constexpr int GB_MAX_VALS = 8;

// per-slot combine for the value table (HF_AGG_*); LDS and global forms use
// hardware ds_*_f64 / global_atomic_*_f64
template <int AOP>
__device__ __forceinline__ void lds_slot_agg(double* a, double v) {
  if (AOP == HF_AGG_SUM)
    unsafeAtomicAdd(a, v);
  else if (AOP == HF_AGG_MIN)
    __hip_atomic_fetch_min(a, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_WORKGROUP);
  else
    __hip_atomic_fetch_max(a, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_WORKGROUP);
}
template <int AOP>
__device__ __forceinline__ void glob_slot_agg(double* a, double v) {
  if (AOP == HF_AGG_SUM)
    unsafeAtomicAdd(a, v);
  else if (AOP == HF_AGG_MIN)
    __hip_atomic_fetch_min(a, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  else
    __hip_atomic_fetch_max(a, v, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}
template <int AOP>
__device__ __forceinline__ double agg_identity() {
  return AOP == HF_AGG_SUM ? 0.0
         : AOP == HF_AGG_MIN ? __longlong_as_double(0x7FF0000000000000LL)
                             : __longlong_as_double(0xFFF0000000000000LL);
}

struct GbPtrs {
  const double* vals[GB_MAX_VALS];
};

template <int NVALS, bool COUNTS, int AOP>
__global__ void __launch_bounds__(BLOCK) k_gb_accum(
    const int64_t* __restrict__ keys, GbPtrs ptrs, int64_t n,
    int64_t key_min, int64_t n_slots,
    double* __restrict__ sums,            // [NVALS][n_slots]
    unsigned long long* __restrict__ rowcnt,  // [n_slots]
    unsigned long long* __restrict__ counts,  // [NVALS][n_slots] or null
    unsigned long long* __restrict__ err) {
  const int64_t npair = n >> 1;
  const longlong2* keys2 = reinterpret_cast<const longlong2*>(keys);
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < npair; i += stride) {
    longlong2 kk = keys2[i];
    const int64_t k0 = kk.x - key_min, k1 = kk.y - key_min;
    const bool ok0 = (uint64_t)k0 < (uint64_t)n_slots;
    const bool ok1 = (uint64_t)k1 < (uint64_t)n_slots;
    if (!ok0 || !ok1) atomicAdd(err, (unsigned long long)(!ok0 + !ok1));
    if (ok0) atomicAdd(&rowcnt[k0], 1ULL);
    if (ok1) atomicAdd(&rowcnt[k1], 1ULL);
#pragma unroll
    for (int c = 0; c < NVALS; ++c) {
      const double2 v = reinterpret_cast<const double2*>(ptrs.vals[c])[i];
      if (ok0 && v.x == v.x) {
        glob_slot_agg<AOP>(&sums[(int64_t)c * n_slots + k0], v.x);
        if (COUNTS) atomicAdd(&counts[(int64_t)c * n_slots + k0], 1ULL);
      }
      if (ok1 && v.y == v.y) {
        glob_slot_agg<AOP>(&sums[(int64_t)c * n_slots + k1], v.y);
        if (COUNTS) atomicAdd(&counts[(int64_t)c * n_slots + k1], 1ULL);
      }
    }
  }
  if ((n & 1) && blockIdx.x == 0 && threadIdx.x == 0) {
    const int64_t k = keys[n - 1] - key_min;
    if ((uint64_t)k < (uint64_t)n_slots) {
      atomicAdd(&rowcnt[k], 1ULL);
      for (int c = 0; c < NVALS; ++c) {
        const double v = ptrs.vals[c][n - 1];
        if (v == v) {
          glob_slot_agg<AOP>(&sums[(int64_t)c * n_slots + k], v);
          if (COUNTS) atomicAdd(&counts[(int64_t)c * n_slots + k], 1ULL);
        }
      }
    } else {
      atomicAdd(err, 1ULL);
    }
  }
}

// ---------------------------------------------------------------------------
// Radix groupby path (the north-star kernel set).
//
// Global f64/u64 atomics saturate at ~23-27 G ops/s on gfx950 (measured,
// tools/atomic_probe.hip; privatizing tables per XCD changes nothing), while
// LDS ds_add_f64 aggregation runs within 14% of the streaming ceiling.  So
// for key ranges beyond LDS capacity the kernel radix-partitions rows into
// buckets of GB_RANGE=8192 keys (u16 local key + f64 value, packed SoA),
// then aggregates each bucket in an LDS-resident dense table:
//   P0 k_gb_hist       : per-bucket row counts (8 B/row read; cached per key
//                        column — immutable columns make this the device
//                        analog of the reference's lazy metadata caches)
//   P1 k_gb_scatter    : read 12 B/row (cached u32 key + f64 val), write
//                        8 B val + 2 B lowkey into exact per-bucket regions
//                        (LDS ranks, one global cursor atomic per
//                        block-tile per bucket); software-pipelined across
//                        tiles (round 2)
//   P2 k_gb_bucket_agg : stream each bucket chunk into an LDS table
//                        (ds_add_f64 behind a register run-accumulator for
//                        skewed keys), merge non-empty slots into the
//                        global dense table (atomics only per slot)
// n_slots <= GB_RANGE skips P0/P1 entirely (k_gb_dense: one 16 B/row pass).
// ---------------------------------------------------------------------------

constexpr int GB_RANGE_LOG = 12;               // 4096 keys per bucket: the
                                               // P2 LDS table is 48 KB -> 3
                                               // blocks/CU (probe: RL12 agg
                                               // 5.0 ms vs RL13 7.2)
constexpr int GB_RANGE = 1 << GB_RANGE_LOG;
constexpr int64_t GB_MAX_BUCKETS = 1024;       // radix path: n_slots <= 4.2M
constexpr int64_t GB_DENSE_MAX = 8192;         // direct-LDS path bound
constexpr int64_t AGG_CHUNK = 1 << 21;         // rows per P2 work item

struct GbWorkItem {
  int64_t start;   // absolute row in the scatter regions
  int32_t bucket;
  int32_t len;
};

__global__ void __launch_bounds__(BLOCK) k_gb_hist(
    const int64_t* __restrict__ keys, int64_t n, int64_t key_min,
    int64_t n_slots, int nb, int range_log,
    unsigned long long* __restrict__ hist) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  unsigned* lhist = reinterpret_cast<unsigned*>(smem_raw);
  for (int t = threadIdx.x; t < nb; t += blockDim.x) lhist[t] = 0;
  __syncthreads();
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int64_t k = keys[i] - key_min;
    if ((uint64_t)k < (uint64_t)n_slots)
      atomicAdd(&lhist[k >> range_log], 1u);
  }
  __syncthreads();
  for (int t = threadIdx.x; t < nb; t += blockDim.x)
    if (lhist[t]) atomicAdd(&hist[t], (unsigned long long)lhist[t]);
}

// P1: LDS-staged bucket-sorted tiles.  Row pairs load 16 B-vectorized
// (longlong2/double2), are ranked into a per-tile LDS histogram (`ds_add`),
// prefix-scanned wave-parallel, staged bucket-SORTED in LDS, and written out
// coalesced — each tile emits one contiguous chunk per bucket stream
// (probe: 2.6x the register-staged direct scatter).  RPT rows/thread
// (even); NV value columns; odd tail row handled by block 0 up front.
__global__ void __launch_bounds__(BLOCK) k_conv_keys32(
    const int64_t* __restrict__ keys, int64_t n, int64_t key_min,
    unsigned* __restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int64_t k = keys[i] - key_min;
    out[i] = ((uint64_t)k < 0xFFFFFFFFull) ? (unsigned)k : 0xFFFFFFFFu;
  }
}

// Software-pipelined tile loop (round-2 probe winner, tools/ring_probe.hip:
// P1 6.69 -> 5.52 ms on the 1e9-row north star).  Raw global loads for tile
// t+1 are issued right after tile t's LDS staging — the registers are dead at
// that point — so their ~700-cycle latency hides under t's writeout + barrier
// instead of serializing at the top of the next iteration (the 81%
// wave-parked stall of the round-1 kernel, profiles/r01b/sq_scatter.txt).
// row0/row1 bound the input rows (the overlapped-P2 path scatters in
// chunks; row0 is always even, only the LAST chunk may end odd)
template <int NV, int RPT, int BLK, int RL, bool K32>
__global__ void __launch_bounds__(BLK) k_gb_scatter(
    const int64_t* __restrict__ keys, const unsigned* __restrict__ keys32,
    const double* __restrict__ v0,
    const double* __restrict__ v1, int64_t row0, int64_t row1,
    int64_t key_min, int64_t n_slots,
    int nb, unsigned* __restrict__ cursors,
    double* __restrict__ r0, double* __restrict__ r1,
    unsigned short* __restrict__ rk, unsigned long long* __restrict__ err) {
  constexpr int TILE = BLK * RPT;
  constexpr int PAIRS = RPT / 2;
  if ((row1 & 1) && blockIdx.x == 0 && threadIdx.x == 0) {
    // odd tail row: direct single-row reservation + write
    const int64_t k =
        K32 ? (int64_t)keys32[row1 - 1] : keys[row1 - 1] - key_min;
    if ((uint64_t)k < (uint64_t)n_slots) {
      const int b = (int)(k >> RL);
      const int64_t pos = (int64_t)atomicAdd(&cursors[b], 1u);
      rk[pos] = (unsigned short)(k & ((1 << RL) - 1));
      if (NV > 0) r0[pos] = v0[row1 - 1];
      if (NV > 1) r1[pos] = v1[row1 - 1];
    } else {
      atomicAdd(err, 1ULL);
    }
  }
  const int64_t pr0 = row0 >> 1;
  const int64_t npair_total = ((row1 & ~1LL) - row0) >> 1;
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  double* sval0 = reinterpret_cast<double*>(smem_raw);            // [TILE]
  double* sval1 = sval0 + (NV > 1 ? TILE : 0);
  unsigned* skey = reinterpret_cast<unsigned*>(
      sval0 + (NV > 1 ? 2 * (int64_t)TILE : (NV > 0 ? TILE : 0)));  // [TILE]
  unsigned* it_cnt = skey + TILE;                                 // [nb]
  unsigned* it_off = it_cnt + nb;                                 // [nb]
  unsigned* it_gbase = it_off + nb;                               // [nb]
  unsigned* s_total = it_gbase + nb;  // scalar; keep ALL LDS in the dynamic
                                      // region (a static __shared__ would
                                      // shift the base off 16B — G17)
  const int64_t ntiles = (npair_total * 2 + TILE - 1) / TILE;

  // only the K32-selected key array and the first NV value arrays are ever
  // touched; the others constant-fold away (template params) and cost no
  // registers
  uint2 kraw[PAIRS];
  longlong2 kraw64[PAIRS];
  double2 vraw0[PAIRS];
  double2 vraw1[PAIRS];
  auto issue_loads = [&](int64_t tile) {
#pragma unroll
    for (int j = 0; j < PAIRS; ++j) {
      const int64_t pj = tile * (TILE / 2) + (int64_t)j * BLK + threadIdx.x;
      if (pj < npair_total) {
        const int64_t pr = pr0 + pj;
        if (K32)
          kraw[j] = reinterpret_cast<const uint2*>(keys32)[pr];
        else
          kraw64[j] = reinterpret_cast<const longlong2*>(keys)[pr];
        if (NV > 0) vraw0[j] = reinterpret_cast<const double2*>(v0)[pr];
        if (NV > 1) vraw1[j] = reinterpret_cast<const double2*>(v1)[pr];
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
    __syncthreads();  // it_cnt cleared (prologue or previous iteration)
#pragma unroll
    for (int j = 0; j < PAIRS; ++j) {
      const int64_t pj = tile * (TILE / 2) + (int64_t)j * BLK + threadIdx.x;
      const int a = 2 * j, bslot = 2 * j + 1;
      lb[a] = lb[bslot] = -1;
      if (pj < npair_total) {
        const int64_t ka = K32 ? (int64_t)kraw[j].x : kraw64[j].x - key_min;
        const int64_t kb = K32 ? (int64_t)kraw[j].y : kraw64[j].y - key_min;
        if ((uint64_t)ka < (uint64_t)n_slots) {
          lb[a] = (int)(ka >> RL);
          lk[a] = (unsigned)(ka & ((1 << RL) - 1));
        } else {
          atomicAdd(err, 1ULL);
        }
        if ((uint64_t)kb < (uint64_t)n_slots) {
          lb[bslot] = (int)(kb >> RL);
          lk[bslot] = (unsigned)(kb & ((1 << RL) - 1));
        } else {
          atomicAdd(err, 1ULL);
        }
      }
    }
#pragma unroll
    for (int j = 0; j < RPT; ++j)
      if (lb[j] >= 0) lr[j] = atomicAdd(&it_cnt[lb[j]], 1u);
    __syncthreads();
    // wave 0: parallel exclusive scan over the nb tile counters
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
        skey[p] = ((unsigned)lb[j] << 16) | lk[j];
        if (NV > 0) sval0[p] = (j & 1) ? vraw0[j >> 1].y : vraw0[j >> 1].x;
        if (NV > 1) sval1[p] = (j & 1) ? vraw1[j >> 1].y : vraw1[j >> 1].x;
      }
    }
    __syncthreads();  // staging visible; it_cnt reads (scan) long done
    // registers are dead — issue the NEXT tile's loads so they fly during
    // this tile's writeout, and clear it_cnt for the next rank phase
    const int64_t nxt = tile + gridDim.x;
    if (nxt < ntiles) issue_loads(nxt);
    for (int t = threadIdx.x; t < nb; t += blockDim.x) it_cnt[t] = 0;
    const int staged = (int)*s_total;
    for (int p = threadIdx.x; p < staged; p += blockDim.x) {
      const unsigned b = skey[p] >> 16;
      const int64_t pos = (int64_t)it_gbase[b] + (p - it_off[b]);
      rk[pos] = (unsigned short)(skey[p] & 0xFFFF);
      if (NV > 0) r0[pos] = sval0[p];
      if (NV > 1) r1[pos] = sval1[p];
    }
    // loop-top barrier orders the writeout's LDS reads against the next
    // tile's staging writes
  }
}

// Build P2 work items ON DEVICE from cursor snapshots (the overlapped
// path): split every bucket's fresh region slice [lo[b], hi[b]) into
// <= max_len-row items.  One block; item order is irrelevant.
__global__ void k_gb_make_work(const unsigned* __restrict__ lo,
                               const unsigned* __restrict__ hi, int nb,
                               int max_len, GbWorkItem* __restrict__ out,
                               int* __restrict__ count) {
  if (blockIdx.x != 0) return;
  __shared__ int s_cnt;
  if (threadIdx.x == 0) s_cnt = 0;
  __syncthreads();
  for (int b = threadIdx.x; b < nb; b += blockDim.x) {
    int64_t s0 = lo[b];
    const int64_t s1 = hi[b];
    while (s0 < s1) {
      const int len = (int)min((int64_t)max_len, s1 - s0);
      const int i = atomicAdd(&s_cnt, 1);
      out[i] = GbWorkItem{s0, b, len};
      s0 += len;
    }
  }
  __syncthreads();
  if (threadIdx.x == 0) *count = s_cnt;
}

// P2: one work item per (bucket, chunk); one value column per launch.
// ROWCNT: also mark group presence (first column / keys-only pass).  The
// global rowcnt table is PRESENCE-ONLY everywhere (compaction tests >0;
// per-column counts live in the counts table), so the per-row mark is a
// plain LDS byte store (last-writer-wins, LDS is CU-local) — no atomic.
// CNT: also merge per-slot non-NaN counts for this column.
// 2 rows/lane vectorized (regions are 64-row aligned; odd chunk tails are
// only ever the last chunk of a bucket).
// n_work (optional): device item count — the overlapped path launches an
// upper-bound grid and builds the items on device from cursor snapshots
// (k_gb_make_work), so extra blocks exit on the count.  Item starts may
// have any parity there: scalar head/tail rows keep the 2-row vector
// middle aligned.
template <bool ROWCNT, bool CNT, bool HAVE_VAL, int RL, int AOP>
__global__ void __launch_bounds__(512) k_gb_bucket_agg(
    const double* __restrict__ vals, const unsigned short* __restrict__ lowkeys,
    const GbWorkItem* __restrict__ work, const int* __restrict__ n_work,
    int64_t n_slots,
    double* __restrict__ gsums, unsigned long long* __restrict__ growcnt,
    unsigned long long* __restrict__ gcounts) {
  constexpr int RANGE = 1 << RL;
  __shared__ double lsums[HAVE_VAL ? RANGE : 1];
  __shared__ unsigned lcnt[CNT ? RANGE : 1];
  __shared__ unsigned char ltouch[RANGE];
  if (n_work && blockIdx.x >= (unsigned)*n_work) return;
  const GbWorkItem w = work[blockIdx.x];
  for (int s = threadIdx.x; s < RANGE; s += blockDim.x) {
    if (HAVE_VAL) lsums[s] = agg_identity<AOP>();
    ltouch[s] = 0;
    if (CNT) lcnt[s] = 0;
  }
  __syncthreads();
  const int64_t end = w.start + w.len;
  auto one_row = [&](int64_t i) {
    const int slot = lowkeys[i];
    ltouch[slot] = 1;
    if (HAVE_VAL) {
      const double v = vals[i];
      if (v == v) {
        lds_slot_agg<AOP>(&lsums[slot], v);
        if (CNT) atomicAdd(&lcnt[slot], 1u);
      }
    }
  };
  int64_t a0 = w.start;
  if ((a0 & 1) && threadIdx.x == 0) one_row(a0);
  a0 += (a0 & 1);
  const int64_t npair = (end - a0) >> 1;
  const ushort2* k2 = reinterpret_cast<const ushort2*>(lowkeys + a0);
  const double2* v2 = reinterpret_cast<const double2*>(vals + a0);
  // register run-accumulation: consecutive rows of a lane's stream that
  // share a slot combine in a register before ONE LDS op — on skewed
  // keys the head slot dominates its bucket and the per-slot ds_add
  // would serialize (zipf P2 was 3.8x the uniform cost); on uniform the
  // run length is ~1 and this is a predictable branch
  int rslot = -1;
  double racc = 0.0;
  unsigned rcnt = 0;
  auto flush = [&]() {
    if (rslot >= 0) {
      lds_slot_agg<AOP>(&lsums[rslot], racc);
      if (CNT) atomicAdd(&lcnt[rslot], rcnt);
    }
    rslot = -1;
    rcnt = 0;
  };
  auto feed = [&](int slot, double v) {
    ltouch[slot] = 1;
    if (HAVE_VAL && v == v) {
      if (slot == rslot) {
        racc = AOP == HF_AGG_SUM ? racc + v
               : AOP == HF_AGG_MIN ? fmin(racc, v) : fmax(racc, v);
        ++rcnt;
      } else {
        flush();
        rslot = slot;
        racc = v;
        rcnt = 1;
      }
    }
  };
  for (int64_t i = threadIdx.x; i < npair; i += blockDim.x) {
    const ushort2 kk = k2[i];
    if (HAVE_VAL) {
      const double2 vv = v2[i];
      feed(kk.x, vv.x);
      feed(kk.y, vv.y);
    } else {
      ltouch[kk.x] = 1;
      ltouch[kk.y] = 1;
    }
  }
  flush();
  if (((end - a0) & 1) && threadIdx.x == 0) one_row(end - 1);
  __syncthreads();
  const int64_t gbase = (int64_t)w.bucket << RL;
  for (int s = threadIdx.x; s < RANGE; s += blockDim.x) {
    if (!ltouch[s] || gbase + s >= n_slots) continue;
    if (HAVE_VAL) glob_slot_agg<AOP>(&gsums[gbase + s], lsums[s]);
    if (ROWCNT) atomicAdd(&growcnt[gbase + s], 1ULL);
    if (CNT) atomicAdd(&gcounts[gbase + s], (unsigned long long)lcnt[s]);
  }
}

// Dense direct path: n_slots <= GB_RANGE, one 16 B/row pass per column.
template <bool ROWCNT, bool CNT, bool HAVE_VAL, int AOP>
__global__ void __launch_bounds__(512) k_gb_dense(
    const int64_t* __restrict__ keys, const double* __restrict__ vals,
    int64_t n, int64_t key_min, int64_t n_slots,
    double* __restrict__ gsums, unsigned long long* __restrict__ growcnt,
    unsigned long long* __restrict__ gcounts,
    unsigned long long* __restrict__ err) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  double* lsums = reinterpret_cast<double*>(smem_raw);       // [n_slots]
  unsigned* lcnt = reinterpret_cast<unsigned*>(
      lsums + (HAVE_VAL ? n_slots : 0));                     // when CNT
  unsigned char* ltouch = reinterpret_cast<unsigned char*>(
      lcnt + (CNT ? n_slots : 0));                           // presence bytes
  for (int64_t s = threadIdx.x; s < n_slots; s += blockDim.x) {
    if (HAVE_VAL) lsums[s] = agg_identity<AOP>();
    ltouch[s] = 0;
    if (CNT) lcnt[s] = 0;
  }
  __syncthreads();
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int64_t k = keys[i] - key_min;
    if ((uint64_t)k >= (uint64_t)n_slots) {
      atomicAdd(err, 1ULL);
      continue;
    }
    ltouch[k] = 1;  // presence-only (plain LDS byte store, CU-local)
    if (HAVE_VAL) {
      const double v = vals[i];
      if (v == v) {
        lds_slot_agg<AOP>(&lsums[k], v);
        if (CNT) atomicAdd(&lcnt[k], 1u);
      }
    }
  }
  __syncthreads();
  for (int64_t s = threadIdx.x; s < n_slots; s += blockDim.x) {
    if (!ltouch[s]) continue;
    if (HAVE_VAL) glob_slot_agg<AOP>(&gsums[s], lsums[s]);
    if (ROWCNT) atomicAdd(&growcnt[s], 1ULL);
    if (CNT) atomicAdd(&gcounts[s], (unsigned long long)lcnt[s]);
  }
}

// ---- hash-table groupby accumulate (unbounded key ranges) ----

__device__ __forceinline__ uint64_t hash_mix64(uint64_t x) {
  // splitmix64 finalizer
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

constexpr long long HASH_EMPTY = 0x8000000000000000LL;  // INT64_MIN sentinel

template <int NVALS, bool COUNTS, int AOP>
__global__ void __launch_bounds__(BLOCK) k_gb_hash_accum(
    const int64_t* __restrict__ keys, GbPtrs ptrs, int64_t n, int64_t H,
    long long* __restrict__ tkey, double* __restrict__ sums,
    unsigned long long* __restrict__ rowcnt,
    unsigned long long* __restrict__ counts,
    unsigned long long* __restrict__ err_full) {
  const int64_t stride_len = H + 1;  // +1: the INT64_MIN special slot
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const long long k = keys[i];
    int64_t slot;
    if (k == HASH_EMPTY) {
      slot = H;
    } else {
      uint64_t h = hash_mix64((uint64_t)k) & (uint64_t)(H - 1);
      slot = -1;
      // bounded sweep: a chain this long means the table is effectively
      // full — fail fast so the host can grow and retry
      const int64_t maxprobe = H < 4096 ? H : 4096;
      for (int64_t probes = 0; probes < maxprobe; ++probes) {
        const long long cur = (long long)atomicCAS(
            (unsigned long long*)&tkey[h], (unsigned long long)HASH_EMPTY,
            (unsigned long long)k);
        if (cur == HASH_EMPTY || cur == k) {
          slot = (int64_t)h;
          break;
        }
        h = (h + 1) & (uint64_t)(H - 1);
      }
      if (slot < 0) {
        atomicAdd(err_full, 1ULL);
        continue;
      }
    }
    atomicAdd(&rowcnt[slot], 1ULL);
#pragma unroll
    for (int c = 0; c < NVALS; ++c) {
      const double v = ptrs.vals[c][i];
      if (v == v) {
        glob_slot_agg<AOP>(&sums[(int64_t)c * stride_len + slot], v);
        if (COUNTS) atomicAdd(&counts[(int64_t)c * stride_len + slot], 1ULL);
      }
    }
  }
}

// Deterministic counter-based RNG fills (bench/test data generation on
// device — bench.py's synthetic frames are born in HBM so the bench is GPU
// work, not host numpy; oracle.rand_* mirrors these formulas bit-exactly
// for the verify gate).  splitmix64 finalizer over (seed + index).
__device__ __forceinline__ uint64_t rng_mix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ull;
  x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
  x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
  return x ^ (x >> 31);
}

__global__ void __launch_bounds__(BLOCK) k_fill_randint(
    int64_t* __restrict__ p, int64_t n, uint64_t seed, int64_t lo,
    uint64_t span) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride)
    p[i] = lo + (int64_t)(rng_mix64(seed + (uint64_t)i) % span);
}

__global__ void __launch_bounds__(BLOCK) k_fill_randf64(
    double* __restrict__ p, int64_t n, uint64_t seed) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride)
    p[i] = (double)(rng_mix64(seed + (uint64_t)i) >> 11) *
           (1.0 / 9007199254740992.0);
}

// inverse-CDF draw (zipf and friends): u uniform in [0,1), output =
// np.searchsorted(cdf, u, side='right') on a sorted f64 cdf column
__global__ void __launch_bounds__(BLOCK) k_fill_randcdf(
    int64_t* __restrict__ p, int64_t n, uint64_t seed,
    const double* __restrict__ cdf, int64_t m) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const double u = (double)(rng_mix64(seed + (uint64_t)i) >> 11) *
                     (1.0 / 9007199254740992.0);
    int64_t a = 0, b = m;          // first idx with cdf[idx] > u
    while (a < b) {
      const int64_t mid = (a + b) >> 1;
      if (cdf[mid] <= u) a = mid + 1; else b = mid;
    }
    p[i] = a;
  }
}

__global__ void __launch_bounds__(BLOCK) k_i64_to_u32(
    const int64_t* __restrict__ in, unsigned* __restrict__ out, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) out[i] = (unsigned)in[i];
}

__global__ void __launch_bounds__(BLOCK) k_fill_i64(int64_t* __restrict__ p,
                                                    int64_t v, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) p[i] = v;
}

// ---- compaction: present slots (rowcnt>0) -> ascending keys + columns ----
// Fixed tile partitioning so prefix order == slot order.
constexpr int COMPACT_TILE = 4096;  // slots per tile, one 256-thread block/tile

__global__ void __launch_bounds__(BLOCK) k_compact_count(
    const unsigned long long* __restrict__ rowcnt, int64_t n_slots,
    int64_t* __restrict__ tile_counts) {
  const int64_t t0 = (int64_t)blockIdx.x * COMPACT_TILE;
  const int64_t t1 = min(t0 + (int64_t)COMPACT_TILE, n_slots);
  int local = 0;
  for (int64_t s = t0 + threadIdx.x; s < t1; s += blockDim.x)
    local += (rowcnt[s] != 0);
  // block reduce
  for (int off = 32; off > 0; off >>= 1) local += __shfl_down(local, off);
  __shared__ int s_c[BLOCK / 64];
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) s_c[wave] = local;
  __syncthreads();
  if (threadIdx.x == 0) {
    int tot = 0;
    for (int w = 0; w < BLOCK / 64; ++w) tot += s_c[w];
    tile_counts[blockIdx.x] = tot;
  }
}

// single-workgroup exclusive scan of tile_counts (ntiles <= 1M/4096*32 ~ small)
__global__ void __launch_bounds__(1024) k_compact_scan(
    int64_t* __restrict__ tile_counts, int64_t ntiles, int64_t* __restrict__ total) {
  __shared__ int64_t carry;
  if (threadIdx.x == 0) carry = 0;
  __syncthreads();
  __shared__ int64_t buf[1024];
  for (int64_t base = 0; base < ntiles; base += 1024) {
    const int64_t i = base + threadIdx.x;
    int64_t v = (i < ntiles) ? tile_counts[i] : 0;
    // Hillis–Steele inclusive scan in LDS
    buf[threadIdx.x] = v;
    __syncthreads();
    for (int off = 1; off < 1024; off <<= 1) {
      int64_t add = (threadIdx.x >= off) ? buf[threadIdx.x - off] : 0;
      __syncthreads();
      buf[threadIdx.x] += add;
      __syncthreads();
    }
    const int64_t incl = buf[threadIdx.x];
    if (i < ntiles) tile_counts[i] = carry + incl - v;  // exclusive
    __syncthreads();
    if (threadIdx.x == 1023) carry += incl;
    __syncthreads();
  }
  if (threadIdx.x == 0) *total = carry;
}

template <bool COUNTS>
__global__ void __launch_bounds__(BLOCK) k_compact_scatter(
    const double* __restrict__ sums, const unsigned long long* __restrict__ rowcnt,
    const unsigned long long* __restrict__ counts, int nvals,
    int64_t key_min, int64_t n_slots,
    const int64_t* __restrict__ tile_offsets,
    int64_t* __restrict__ out_keys, double* const* __restrict__ out_sums,
    int64_t* const* __restrict__ out_counts) {
  const int64_t t0 = (int64_t)blockIdx.x * COMPACT_TILE;
  const int64_t t1 = min(t0 + (int64_t)COMPACT_TILE, n_slots);
  __shared__ int64_t s_base;
  __shared__ int s_wave_cnt[BLOCK / 64];
  if (threadIdx.x == 0) s_base = tile_offsets[blockIdx.x];
  __syncthreads();
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  for (int64_t chunk = t0; chunk < t1; chunk += blockDim.x) {
    const int64_t s = chunk + threadIdx.x;
    const bool pred = (s < t1) && (rowcnt[s] != 0);
    const uint64_t ballot = __ballot(pred);
    if (lane == 0) s_wave_cnt[wave] = __popcll(ballot);
    __syncthreads();
    int64_t wave_base = 0;
    for (int w = 0; w < wave; ++w) wave_base += s_wave_cnt[w];
    if (pred) {
      const int64_t pos = s_base + wave_base +
          __popcll(ballot & ((lane == 63) ? ~0ULL >> 1 : ((1ULL << lane) - 1)));
      out_keys[pos] = key_min + s;
      for (int c = 0; c < nvals; ++c) {
        out_sums[c][pos] = sums[(int64_t)c * n_slots + s];
        if (COUNTS) out_counts[c][pos] = (int64_t)counts[(int64_t)c * n_slots + s];
      }
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      int64_t tot = 0;
      for (int w = 0; w < BLOCK / 64; ++w) tot += s_wave_cnt[w];
      s_base += tot;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// Join kernels — broadcast-right dense-range CSR inner join
// (device form of MergeImpl.row_axis_merge, merge.py:104-178; see
// include/hipframe.h).  Build: count -> scan -> fill -> order fixup.
// Probe: two passes so the output preserves pandas' match order exactly
// (left row order, right row order within key): pass A stores per-left-row
// (offset, count) and per-tile totals; a scan of tile totals gives exact
// output positions; pass B emits.
// ---------------------------------------------------------------------------

constexpr int JOIN_TILE = 4096;           // left rows per probe tile
constexpr int JOIN_MAX_DUP = 4096;        // right rows per key cap (fixup sort)

__global__ void __launch_bounds__(BLOCK) k_hist_u32(
    const int64_t* __restrict__ keys, int64_t n, int64_t key_min,
    int64_t n_slots, unsigned* __restrict__ cnt,
    unsigned long long* __restrict__ err) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int64_t k = keys[i] - key_min;
    if ((uint64_t)k < (uint64_t)n_slots)
      atomicAdd(&cnt[k], 1u);
    else
      atomicAdd(err, 1ULL);
  }
}

// tile sums of a u32 array (for the big exclusive scan building the CSR)
__global__ void __launch_bounds__(BLOCK) k_tile_sums_u32(
    const unsigned* __restrict__ v, int64_t n, int64_t* __restrict__ sums) {
  const int64_t t0 = (int64_t)blockIdx.x * JOIN_TILE;
  const int64_t t1 = min(t0 + (int64_t)JOIN_TILE, n);
  long long local = 0;
  for (int64_t s = t0 + threadIdx.x; s < t1; s += blockDim.x) local += v[s];
  for (int off = 32; off > 0; off >>= 1) local += __shfl_down(local, off);
  __shared__ long long sc[BLOCK / 64];
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) sc[wave] = local;
  __syncthreads();
  if (threadIdx.x == 0) {
    long long tot = 0;
    for (int w = 0; w < BLOCK / 64; ++w) tot += sc[w];
    sums[blockIdx.x] = tot;
  }
}

// apply tile bases: csr[s] = tile_base + local exclusive prefix; also writes
// csr[n] = grand total (u64 offsets)
__global__ void __launch_bounds__(BLOCK) k_scan_apply_u32(
    const unsigned* __restrict__ v, int64_t n,
    const int64_t* __restrict__ tile_bases, const int64_t* __restrict__ total,
    unsigned long long* __restrict__ csr) {
  __shared__ unsigned long long s_run;
  const int64_t t0 = (int64_t)blockIdx.x * JOIN_TILE;
  const int64_t t1 = min(t0 + (int64_t)JOIN_TILE, n);
  if (threadIdx.x == 0) s_run = (unsigned long long)tile_bases[blockIdx.x];
  __shared__ int s_wave_sum[BLOCK / 64];
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  __syncthreads();
  for (int64_t chunk = t0; chunk < t1; chunk += blockDim.x) {
    const int64_t s = chunk + threadIdx.x;
    const unsigned x = (s < t1) ? v[s] : 0;
    // wave-inclusive scan
    unsigned incl = x;
    for (int d = 1; d < 64; d <<= 1) {
      unsigned up = __shfl_up(incl, d);
      if (lane >= d) incl += up;
    }
    if (lane == 63) s_wave_sum[wave] = (int)incl;
    __syncthreads();
    unsigned wave_base = 0;
    for (int w = 0; w < wave; ++w) wave_base += (unsigned)s_wave_sum[w];
    if (s < t1) csr[s] = s_run + wave_base + incl - x;
    __syncthreads();
    if (threadIdx.x == 0) {
      unsigned tot = 0;
      for (int w = 0; w < BLOCK / 64; ++w) tot += (unsigned)s_wave_sum[w];
      s_run += tot;
    }
    __syncthreads();
  }
  if (blockIdx.x == 0 && threadIdx.x == 0) csr[n] = (unsigned long long)*total;
}

struct JoinPtrs {
  double* vals[GB_MAX_VALS];
};
struct JoinConstPtrs {
  const double* vals[GB_MAX_VALS];
};

template <int NR>
__global__ void __launch_bounds__(BLOCK) k_join_fill(
    const int64_t* __restrict__ keys, JoinConstPtrs rv, int64_t n,
    int64_t key_min, int64_t n_slots, const unsigned long long* __restrict__ csr,
    unsigned* __restrict__ cursor, unsigned* __restrict__ jidx, JoinPtrs jv) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int64_t k = keys[i] - key_min;
    if ((uint64_t)k >= (uint64_t)n_slots) continue;  // counted in hist err
    const unsigned r = atomicAdd(&cursor[k], 1u);
    const int64_t pos = (int64_t)csr[k] + r;
    jidx[pos] = (unsigned)i;
#pragma unroll
    for (int c = 0; c < NR; ++c) jv.vals[c][pos] = rv.vals[c][i];
  }
}

// restore right-row order within each key (pandas match order): tiny
// insertion sort per multi-occupancy slot, one thread per slot
template <int NR>
__global__ void __launch_bounds__(BLOCK) k_join_fixup(
    const unsigned long long* __restrict__ csr, int64_t n_slots,
    unsigned* __restrict__ jidx, JoinPtrs jv,
    unsigned long long* __restrict__ err) {
  int64_t s = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; s < n_slots; s += stride) {
    const int64_t lo = (int64_t)csr[s], hi = (int64_t)csr[s + 1];
    const int64_t cnt = hi - lo;
    if (cnt <= 1) continue;
    if (cnt > JOIN_MAX_DUP) {
      atomicAdd(err, 1ULL);
      continue;
    }
    for (int64_t a = lo + 1; a < hi; ++a) {
      const unsigned ki = jidx[a];
      double kv[NR > 0 ? NR : 1];
#pragma unroll
      for (int c = 0; c < NR; ++c) kv[c] = jv.vals[c][a];
      int64_t b = a - 1;
      while (b >= lo && jidx[b] > ki) {
        jidx[b + 1] = jidx[b];
#pragma unroll
        for (int c = 0; c < NR; ++c) jv.vals[c][b + 1] = jv.vals[c][b];
        --b;
      }
      jidx[b + 1] = ki;
#pragma unroll
      for (int c = 0; c < NR; ++c) jv.vals[c][b + 1] = kv[c];
    }
  }
}

// probe pass A: per-left-row offset+count, per-tile match totals
__global__ void __launch_bounds__(BLOCK) k_probe_count(
    const int64_t* __restrict__ lkeys, int64_t n, int64_t key_min,
    int64_t n_slots, const unsigned long long* __restrict__ csr,
    unsigned long long* __restrict__ offs, unsigned* __restrict__ cnts,
    int64_t* __restrict__ tile_sums) {
  const int64_t t0 = (int64_t)blockIdx.x * JOIN_TILE;
  const int64_t t1 = min(t0 + (int64_t)JOIN_TILE, n);
  long long local = 0;
  for (int64_t i = t0 + threadIdx.x; i < t1; i += blockDim.x) {
    const int64_t k = lkeys[i] - key_min;
    unsigned c = 0;
    unsigned long long o = 0;
    if ((uint64_t)k < (uint64_t)n_slots) {
      o = csr[k];
      c = (unsigned)(csr[k + 1] - o);
    }
    offs[i] = o;
    cnts[i] = c;
    local += c;
  }
  for (int off = 32; off > 0; off >>= 1) local += __shfl_down(local, off);
  __shared__ long long sc[BLOCK / 64];
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) sc[wave] = local;
  __syncthreads();
  if (threadIdx.x == 0) {
    long long tot = 0;
    for (int w = 0; w < BLOCK / 64; ++w) tot += sc[w];
    tile_sums[blockIdx.x] = tot;
  }
}

// probe pass B: emit matches at exact positions
template <int NR>
__global__ void __launch_bounds__(BLOCK) k_probe_emit(
    const int64_t* __restrict__ lkeys, int64_t n, int64_t lbase,
    const unsigned long long* __restrict__ offs,
    const unsigned* __restrict__ cnts, const int64_t* __restrict__ tile_bases,
    const unsigned* __restrict__ jidx_unused, JoinConstPtrs jv,
    int64_t* __restrict__ out_keys, int64_t* __restrict__ out_lidx,
    JoinPtrs out_r) {
  __shared__ unsigned long long s_run;
  __shared__ int s_wave_sum[BLOCK / 64];
  const int64_t t0 = (int64_t)blockIdx.x * JOIN_TILE;
  const int64_t t1 = min(t0 + (int64_t)JOIN_TILE, n);
  if (threadIdx.x == 0) s_run = (unsigned long long)tile_bases[blockIdx.x];
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  __syncthreads();
  for (int64_t chunk = t0; chunk < t1; chunk += blockDim.x) {
    const int64_t i = chunk + threadIdx.x;
    const unsigned c = (i < t1) ? cnts[i] : 0;
    unsigned incl = c;
    for (int d = 1; d < 64; d <<= 1) {
      unsigned up = __shfl_up(incl, d);
      if (lane >= d) incl += up;
    }
    if (lane == 63) s_wave_sum[wave] = (int)incl;
    __syncthreads();
    unsigned wave_base = 0;
    for (int w = 0; w < wave; ++w) wave_base += (unsigned)s_wave_sum[w];
    if (i < t1 && c) {
      int64_t dst = (int64_t)(s_run + wave_base + incl - c);
      const int64_t k = lkeys[i];
      const unsigned long long o = offs[i];
      for (unsigned j = 0; j < c; ++j) {
        out_keys[dst + j] = k;
        out_lidx[dst + j] = lbase + i;
#pragma unroll
        for (int cc = 0; cc < NR; ++cc)
          out_r.vals[cc][dst + j] = jv.vals[cc][o + j];
      }
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      unsigned tot = 0;
      for (int w = 0; w < BLOCK / 64; ++w) tot += (unsigned)s_wave_sum[w];
      s_run += tot;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// Sort kernels — stable LSD radix over (u32 key, u32 idx) pairs.
// A "tile" is 64 lanes x SORT_RPT rows owned by ONE wave: lanes are in
// order and the per-step leader loop ranks same-digit lanes by lane id, so
// every pass is stable without cross-wave coordination.
// ---------------------------------------------------------------------------

constexpr int SORT_RPT = 32;                 // rows per lane per tile
constexpr int SORT_TILE = 64 * SORT_RPT;     // 2048 rows per wave-tile
constexpr int SORT_WPB = 4;                  // waves per block

__global__ void __launch_bounds__(BLOCK) k_sort_pack(
    const int64_t* __restrict__ keys, int64_t n, int64_t key_min,
    int64_t key_span, int ascending, unsigned long long* __restrict__ pairs) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const unsigned k = (unsigned)(keys[i] - key_min);
    const unsigned kk = ascending ? k : (unsigned)key_span - k;
    pairs[i] = ((unsigned long long)(unsigned)i << 32) | kk;
  }
}

// per-wave-tile digit histogram -> C[tile][256] (row-major, coalesced)
__global__ void __launch_bounds__(BLOCK) k_sort_count(
    const unsigned long long* __restrict__ pairs, int64_t n, int shift,
    unsigned* __restrict__ C, int64_t ntiles) {
  __shared__ unsigned hist[SORT_WPB][256];
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const int64_t wtiles_per_block = SORT_WPB;
  for (int64_t tile = (int64_t)blockIdx.x * wtiles_per_block + wave;
       tile < ntiles; tile += (int64_t)gridDim.x * wtiles_per_block) {
    for (int d = lane; d < 256; d += 64) hist[wave][d] = 0;
    __builtin_amdgcn_wave_barrier();
    const int64_t t0 = tile * SORT_TILE;
#pragma unroll 4
    for (int j = 0; j < SORT_RPT; ++j) {
      const int64_t row = t0 + (int64_t)j * 64 + lane;
      if (row < n) {
        const unsigned d = ((unsigned)pairs[row] >> shift) & 255u;
        atomicAdd(&hist[wave][d], 1u);  // LDS, same-wave only
      }
    }
    __builtin_amdgcn_wave_barrier();
    for (int d = lane; d < 256; d += 64) C[tile * 256 + d] = hist[wave][d];
  }
}

// 32x32 tiled transpose: C[ntiles][256] -> CT[256][ntiles]
__global__ void __launch_bounds__(1024) k_transpose256(
    const unsigned* __restrict__ C, unsigned* __restrict__ CT,
    int64_t ntiles) {
  __shared__ unsigned t[32][33];
  const int64_t tx0 = (int64_t)blockIdx.x * 32;       // tile-index block
  const int dy = (int)(blockIdx.y * 32);              // digit block (0..224)
  const int lx = threadIdx.x & 31, ly = threadIdx.x >> 5;  // 32x32 threads
  const int64_t src_row = tx0 + ly;
  if (src_row < ntiles) t[ly][lx] = C[src_row * 256 + dy + lx];
  __syncthreads();
  const int64_t dst_col = tx0 + lx;
  if (dst_col < ntiles) CT[(int64_t)(dy + ly) * ntiles + dst_col] = t[lx][ly];
}

// stable scatter: each wave re-walks its tile in order; same-digit lanes
// are matched with 8 BIT BALLOTS (round 2 — the old per-distinct-digit
// leader loop ran ~50 serial iterations per 64-row step; bit-ballot cut
// one pass 29.4 -> 17.5 ms at 1e9 rows, profiles/r02/sort_ballot.log):
// lanes sharing this lane's digit = AND over digit bits of
// (bit ? ballot : ~ballot); rank = popcount(same & below); the lowest
// member advances the digit base.
__global__ void __launch_bounds__(BLOCK) k_sort_scatter(
    const unsigned long long* __restrict__ pairs, int64_t n, int shift,
    const unsigned long long* __restrict__ offs,   // [256][ntiles] exclusive
    int64_t ntiles, unsigned long long* __restrict__ out) {
  __shared__ unsigned long long base[SORT_WPB][256];
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  const unsigned long long below = (1ULL << lane) - 1ULL;
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
        if (lane == __ffsll((long long)same) - 1)
          base[wave][d] += __popcll(same);
        out[pos] = p;
      }
      __builtin_amdgcn_wave_barrier();
    }
  }
}

// ---- wide variant: u64 shifted keys + u32 origins as two arrays (for key
// spans beyond 2^27; up to 8 digit passes) ----

__global__ void __launch_bounds__(BLOCK) k_sort_pack_wide(
    const int64_t* __restrict__ keys, int64_t n, int64_t key_min,
    uint64_t key_span, int ascending, unsigned long long* __restrict__ karr,
    unsigned* __restrict__ iarr) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const uint64_t k = (uint64_t)(keys[i] - key_min);
    karr[i] = ascending ? k : key_span - k;
    iarr[i] = (unsigned)i;
  }
}

__global__ void __launch_bounds__(BLOCK) k_sort_count_wide(
    const unsigned long long* __restrict__ karr, int64_t n, int shift,
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
      if (row < n) {
        const unsigned d = (unsigned)(karr[row] >> shift) & 255u;
        atomicAdd(&hist[wave][d], 1u);
      }
    }
    __builtin_amdgcn_wave_barrier();
    for (int d = lane; d < 256; d += 64) C[tile * 256 + d] = hist[wave][d];
  }
}

__global__ void __launch_bounds__(BLOCK) k_sort_scatter_wide(
    const unsigned long long* __restrict__ karr,
    const unsigned* __restrict__ iarr, int64_t n, int shift,
    const unsigned long long* __restrict__ offs, int64_t ntiles,
    unsigned long long* __restrict__ karr_out, unsigned* __restrict__ iarr_out) {
  __shared__ unsigned long long base[SORT_WPB][256];
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  for (int64_t tile = (int64_t)blockIdx.x * SORT_WPB + wave; tile < ntiles;
       tile += (int64_t)gridDim.x * SORT_WPB) {
    for (int d = lane; d < 256; d += 64)
      base[wave][d] = offs[(int64_t)d * ntiles + tile];
    __builtin_amdgcn_wave_barrier();
    const int64_t t0 = tile * SORT_TILE;
    const unsigned long long below = (1ULL << lane) - 1ULL;
    for (int j = 0; j < SORT_RPT; ++j) {
      const int64_t row = t0 + (int64_t)j * 64 + lane;
      const bool valid = row < n;
      const unsigned long long k = valid ? karr[row] : 0;
      const unsigned idx = valid ? iarr[row] : 0;
      const unsigned d = (unsigned)(k >> shift) & 255u;
      // bit-ballot same-digit matching (see k_sort_scatter)
      unsigned long long same = __ballot(valid);
#pragma unroll
      for (int b = 0; b < 8; ++b) {
        const unsigned long long m = __ballot((d >> b) & 1u);
        same &= ((d >> b) & 1u) ? m : ~m;
      }
      if (valid) {
        const unsigned rank = (unsigned)__popcll(same & below);
        const unsigned long long pos = base[wave][d] + rank;
        if (lane == __ffsll((long long)same) - 1)
          base[wave][d] += __popcll(same);
        karr_out[pos] = k;
        iarr_out[pos] = idx;
      }
      __builtin_amdgcn_wave_barrier();
    }
  }
}

__global__ void __launch_bounds__(BLOCK) k_sort_unpack_wide(
    const unsigned* __restrict__ iarr, int64_t n, int64_t* __restrict__ perm) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) perm[i] = (int64_t)iarr[i];
}

__global__ void __launch_bounds__(BLOCK) k_sort_unpack(
    const unsigned long long* __restrict__ pairs, int64_t n,
    int64_t* __restrict__ perm) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) perm[i] = (int64_t)(pairs[i] >> 32);
}

__global__ void __launch_bounds__(BLOCK) k_fill_f64(double* __restrict__ p,
                                                    double v, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) p[i] = v;
}

__global__ void __launch_bounds__(BLOCK) k_fixup_empty(
    const double* __restrict__ val, const int64_t* __restrict__ cnt,
    double* __restrict__ out, int64_t n) {
  const double nanv = __longlong_as_double(0x7FF8000000000000LL);
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) out[i] = cnt[i] ? val[i] : nanv;
}

// ---- compare + filter kernels ----

template <int OP, typename T>
__device__ __forceinline__ long long cmp1(T x, double s) {
  switch (OP) {
    case HF_CMP_GT: return x > s;
    case HF_CMP_GE: return x >= s;
    case HF_CMP_LT: return x < s;
    case HF_CMP_LE: return x <= s;
    case HF_CMP_EQ: return x == s;
    case HF_CMP_NE: return !(x == s);  // NaN != s -> true (pandas)
    case HF_CMP_NOTNA: return x == x;  // int64: always 1
  }
  return 0;
}

template <int OP, typename T>
__global__ void __launch_bounds__(BLOCK) k_compare(const T* __restrict__ in,
                                                   long long* __restrict__ out,
                                                   double s, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) out[i] = cmp1<OP, T>(in[i], s);
}

constexpr int FILT_TILE = 4096;

__global__ void __launch_bounds__(BLOCK) k_filter_count(
    const long long* __restrict__ mask, int64_t n, int64_t* __restrict__ sums) {
  const int64_t t0 = (int64_t)blockIdx.x * FILT_TILE;
  const int64_t t1 = min(t0 + (int64_t)FILT_TILE, n);
  long long local = 0;
  for (int64_t i = t0 + threadIdx.x; i < t1; i += blockDim.x)
    local += (mask[i] != 0);
  for (int off = 32; off > 0; off >>= 1) local += __shfl_down(local, off);
  __shared__ long long sc[BLOCK / 64];
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) sc[wave] = local;
  __syncthreads();
  if (threadIdx.x == 0) {
    long long tot = 0;
    for (int w = 0; w < BLOCK / 64; ++w) tot += sc[w];
    sums[blockIdx.x] = tot;
  }
}

// order-preserving compaction: ballot-ranked within tile (the same pattern
// as k_compact_scatter).  IOTA: emit base + row index instead of data.
template <typename T, bool IOTA>
__global__ void __launch_bounds__(BLOCK) k_filter_scatter(
    const long long* __restrict__ mask, const T* __restrict__ in, int64_t n,
    const int64_t* __restrict__ tile_bases, int64_t iota_base,
    T* __restrict__ out) {
  const int64_t t0 = (int64_t)blockIdx.x * FILT_TILE;
  const int64_t t1 = min(t0 + (int64_t)FILT_TILE, n);
  __shared__ int64_t s_base;
  __shared__ int s_wave_cnt[BLOCK / 64];
  if (threadIdx.x == 0) s_base = tile_bases[blockIdx.x];
  __syncthreads();
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  for (int64_t chunk = t0; chunk < t1; chunk += blockDim.x) {
    const int64_t i = chunk + threadIdx.x;
    const bool pred = (i < t1) && (mask[i] != 0);
    const uint64_t ballot = __ballot(pred);
    if (lane == 0) s_wave_cnt[wave] = __popcll(ballot);
    __syncthreads();
    int64_t wave_base = 0;
    for (int w = 0; w < wave; ++w) wave_base += s_wave_cnt[w];
    if (pred) {
      const int64_t pos = s_base + wave_base +
          __popcll(ballot & ((lane == 63) ? ~0ULL >> 1 : ((1ULL << lane) - 1)));
      out[pos] = IOTA ? (T)(iota_base + i) : in[i];
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      int64_t tot = 0;
      for (int w = 0; w < BLOCK / 64; ++w) tot += s_wave_cnt[w];
      s_base += tot;
    }
    __syncthreads();
  }
}

// ---- sort-based groupby kernels ----

__global__ void __launch_bounds__(BLOCK) k_head_flags(
    const int64_t* __restrict__ k, int64_t n, long long* __restrict__ head) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride)
    head[i] = (i == 0) || (k[i] != k[i - 1]);
}

// per-row run id = (heads at positions <= i) - 1, via the filter plan's
// tile bases; one device atomic per row into the per-run aggregate
template <int AOP, bool CNT, bool ROWCNT, bool HAVE_VAL>
__global__ void __launch_bounds__(BLOCK) k_segagg(
    const long long* __restrict__ head, const double* __restrict__ vals,
    int64_t n, const int64_t* __restrict__ tile_bases,
    double* __restrict__ gsums, unsigned long long* __restrict__ growcnt,
    unsigned long long* __restrict__ gcounts) {
  const int64_t t0 = (int64_t)blockIdx.x * FILT_TILE;
  const int64_t t1 = min(t0 + (int64_t)FILT_TILE, n);
  __shared__ int64_t s_base;
  __shared__ int s_wave_cnt[BLOCK / 64];
  if (threadIdx.x == 0) s_base = tile_bases[blockIdx.x];
  __syncthreads();
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  for (int64_t chunk = t0; chunk < t1; chunk += blockDim.x) {
    const int64_t i = chunk + threadIdx.x;
    const bool pred = (i < t1) && (head[i] != 0);
    const uint64_t ballot = __ballot(pred);
    if (lane == 0) s_wave_cnt[wave] = __popcll(ballot);
    __syncthreads();
    int64_t wave_base = 0;
    for (int w = 0; w < wave; ++w) wave_base += s_wave_cnt[w];
    if (i < t1) {
      // inclusive count of heads <= i within the tile, minus 1 for run id
      const int64_t run =
          s_base + wave_base +
          __popcll(ballot & ((lane == 63) ? ~0ULL : ((2ULL << lane) - 1))) - 1;
      if (ROWCNT) atomicAdd(&growcnt[run], 1ULL);
      if (HAVE_VAL) {
        const double v = vals[i];
        if (v == v) {
          glob_slot_agg<AOP>(&gsums[run], v);
          if (CNT) atomicAdd(&gcounts[run], 1ULL);
        }
      }
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      int64_t tot = 0;
      for (int w = 0; w < BLOCK / 64; ++w) tot += s_wave_cnt[w];
      s_base += tot;
    }
    __syncthreads();
  }
}

template <typename T, int OP>
__device__ __forceinline__ T cs_ident() {
  if (OP == HF_AGG_MIN)
    return (sizeof(T) == 8 && (T)0.5 == 0) ? (T)INT64_MAX
                                           : (T)__builtin_huge_val();
  if (OP == HF_AGG_MAX)
    return (sizeof(T) == 8 && (T)0.5 == 0) ? (T)INT64_MIN
                                           : (T)-__builtin_huge_val();
  if (OP == HF_AGG_PROD) return T(1);
  return T(0);
}
template <typename T, int OP>
__device__ __forceinline__ T cs_comb(T a, T b) {
  if (OP == HF_AGG_MIN) return a < b ? a : b;
  if (OP == HF_AGG_MAX) return a > b ? a : b;
  if (OP == HF_AGG_PROD) return a * b;
  return a + b;
}
template <typename T, int OP>
__device__ __forceinline__ T cs_load(const T* p_, int64_t i) {
  const T v = p_[i];
  return (v != v) ? cs_ident<T, OP>() : v;  // NaN skipped (f64)
}

template <typename T, int OP>
__global__ void __launch_bounds__(BLOCK) k_cumsum_tiles(
    const T* __restrict__ in, int64_t n, T* __restrict__ tile_sums) {
  const int64_t t0 = (int64_t)blockIdx.x * FILT_TILE;
  const int64_t t1 = min(t0 + (int64_t)FILT_TILE, n);
  T local = cs_ident<T, OP>();
  for (int64_t i = t0 + threadIdx.x; i < t1; i += blockDim.x)
    local = cs_comb<T, OP>(local, cs_load<T, OP>(in, i));
  __shared__ T sc[BLOCK / 64];
  for (int off = 32; off > 0; off >>= 1)
    local = cs_comb<T, OP>(local, (T)__shfl_down(local, off));
  const int lane = threadIdx.x & 63, wave = threadIdx.x >> 6;
  if (lane == 0) sc[wave] = local;
  __syncthreads();
  if (threadIdx.x == 0) {
    T tot = cs_ident<T, OP>();
    for (int w = 0; w < BLOCK / 64; ++w) tot = cs_comb<T, OP>(tot, sc[w]);
    tile_sums[blockIdx.x] = tot;
  }
}

template <typename T, int OP>
__global__ void __launch_bounds__(1024) k_cumsum_scan_tiles(
    T* __restrict__ tile_sums, int64_t ntiles) {
  // exclusive scan under cs_comb: track each element's PREFIX (exclusive)
  // explicitly since min/max have no inverse
  __shared__ T carry;
  if (threadIdx.x == 0) carry = cs_ident<T, OP>();
  __syncthreads();
  __shared__ T buf[1024];
  for (int64_t base = 0; base < ntiles; base += 1024) {
    const int64_t i = base + threadIdx.x;
    T v = (i < ntiles) ? tile_sums[i] : cs_ident<T, OP>();
    buf[threadIdx.x] = v;
    __syncthreads();
    T excl = cs_ident<T, OP>();
    for (int off = 1; off < 1024; off <<= 1) {
      T add = (threadIdx.x >= off) ? buf[threadIdx.x - off]
                                   : cs_ident<T, OP>();
      __syncthreads();
      buf[threadIdx.x] = cs_comb<T, OP>(buf[threadIdx.x], add);
      __syncthreads();
    }
    const T incl = buf[threadIdx.x];
    __syncthreads();
    buf[threadIdx.x] = incl;  // reuse buf to read neighbor inclusives
    __syncthreads();
    // exclusive = inclusive of the previous lane (or identity at lane 0)
    excl = (threadIdx.x == 0) ? cs_ident<T, OP>() : buf[threadIdx.x - 1];
    if (i < ntiles) tile_sums[i] = cs_comb<T, OP>(carry, excl);
    __syncthreads();
    if (threadIdx.x == 1023) carry = cs_comb<T, OP>(carry, incl);
    __syncthreads();
  }
}

template <typename T, int OP>
__global__ void __launch_bounds__(BLOCK) k_cumsum_apply(
    const T* __restrict__ in, int64_t n, const T* __restrict__ tile_base,
    T* __restrict__ out) {
  const int64_t t0 = (int64_t)blockIdx.x * FILT_TILE;
  const int64_t t1 = min(t0 + (int64_t)FILT_TILE, n);
  constexpr int PER = FILT_TILE / BLOCK;  // elems per thread
  // stage the tile through LDS so HBM access stays COALESCED (striped)
  // while each thread scans PER CONSECUTIVE elements from LDS
  __shared__ T stage[FILT_TILE];
  for (int j = 0; j < PER; ++j) {
    const int64_t i = t0 + (int64_t)j * BLOCK + threadIdx.x;
    if (i < t1) stage[(int)(i - t0)] = in[i];
  }
  __syncthreads();
  const int s0 = threadIdx.x * PER;
  const int lim = (int)(t1 - t0);
  T loc[PER];
  T run = cs_ident<T, OP>();
  for (int j = 0; j < PER; ++j) {
    if (s0 + j < lim) {
      const T v = stage[s0 + j];
      run = cs_comb<T, OP>(run, (v != v) ? cs_ident<T, OP>() : v);
    }
    loc[j] = run;  // inclusive within the thread's segment
  }
  // exclusive base across threads: scan thread totals, read neighbor
  __shared__ T buf[BLOCK];
  buf[threadIdx.x] = run;
  __syncthreads();
  for (int off = 1; off < BLOCK; off <<= 1) {
    T add = (threadIdx.x >= off) ? buf[threadIdx.x - off]
                                 : cs_ident<T, OP>();
    __syncthreads();
    buf[threadIdx.x] = cs_comb<T, OP>(buf[threadIdx.x], add);
    __syncthreads();
  }
  const T incl = buf[threadIdx.x];
  __syncthreads();
  buf[threadIdx.x] = incl;
  __syncthreads();
  const T texcl = (threadIdx.x == 0) ? cs_ident<T, OP>()
                                     : buf[threadIdx.x - 1];
  const T tbase = cs_comb<T, OP>(tile_base[blockIdx.x], texcl);
  __syncthreads();
  for (int j = 0; j < PER; ++j) {
    if (s0 + j < lim) {
      const T v0 = stage[s0 + j];
      stage[s0 + j] = (v0 != v0) ? v0 : cs_comb<T, OP>(tbase, loc[j]);
    }
  }
  __syncthreads();
  for (int j = 0; j < PER; ++j) {
    const int64_t i = t0 + (int64_t)j * BLOCK + threadIdx.x;
    if (i < t1) out[i] = stage[(int)(i - t0)];
  }
}

__global__ void __launch_bounds__(BLOCK) k_f64_ordered(
    const double* __restrict__ in, int64_t* __restrict__ out, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    double x = in[i] + 0.0;  // -0.0 -> +0.0 (pandas groups them together)
    // canonicalize NaN (any sign/payload) to one ordered value: pandas
    // treats every NaN as equal — groupby keys, sort ties AND merge keys
    // (pandas matches NaN==NaN in merges)
    long long v = (x != x) ? 0x7FF8000000000000LL
                           : __double_as_longlong(x);
    // signed total order: non-negative floats keep their bits (already
    // increasing as signed i64); negative floats reverse below zero
    out[i] = (v < 0) ? (~v ^ 0x8000000000000000LL) : v;
  }
}

__global__ void __launch_bounds__(BLOCK) k_ordered_f64(
    const int64_t* __restrict__ in, double* __restrict__ out, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const long long v = in[i];
    out[i] = __longlong_as_double(
        (v < 0) ? ~(v ^ 0x8000000000000000LL) : v);
  }
}

__global__ void __launch_bounds__(BLOCK) k_search_sorted(
    const int64_t* __restrict__ keys, int64_t n,
    const int64_t* __restrict__ sorted, int64_t m,
    int64_t* __restrict__ out) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int64_t k = keys[i];
    int64_t lo = 0, hi = m;
    while (lo < hi) {  // lower_bound; first levels stay L2-resident
      const int64_t mid = (lo + hi) >> 1;
      if (sorted[mid] < k) lo = mid + 1; else hi = mid;
    }
    out[i] = (lo < m && sorted[lo] == k) ? lo : -1;
  }
}

__global__ void __launch_bounds__(BLOCK) k_shuffle_dest(
    const int64_t* __restrict__ keys, const int64_t* __restrict__ split,
    int nsplit, long long* __restrict__ dest, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    const int64_t k = keys[i];
    int d = 0;
    for (int j = 0; j < nsplit; ++j) d += (split[j] <= k);
    dest[i] = d;
  }
}

__global__ void __launch_bounds__(BLOCK) k_gather_f64(
    const double* __restrict__ src, const int64_t* __restrict__ idx,
    double* __restrict__ out, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) out[i] = src[idx[i]];
}
__global__ void __launch_bounds__(BLOCK) k_gather_i64(
    const int64_t* __restrict__ src, const int64_t* __restrict__ idx,
    int64_t* __restrict__ out, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) out[i] = src[idx[i]];
}

// Cross-join row indices: lidx[i] = i / nr, ridx[i] = i % nr — the gather
// indices materializing a cartesian product (pandas merge how='cross').
__global__ void __launch_bounds__(BLOCK) k_cross_idx(
    int64_t n, int64_t nr, int64_t* __restrict__ lidx,
    int64_t* __restrict__ ridx) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    lidx[i] = i / nr;
    ridx[i] = i - (i / nr) * nr;
  }
}

// Inverse of gather: out[idx[i]] = src[i].  idx must be a permutation of
// [0, n) (the caller holds a sort_perm) so writes never collide; reads are
// coalesced and the 8 B scattered writes coalesce in L2 like the radix
// scatter's do.
template <typename T>
__global__ void __launch_bounds__(BLOCK) k_scatter(
    const T* __restrict__ src, const int64_t* __restrict__ idx,
    T* __restrict__ out, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) out[idx[i]] = src[i];
}

// ---------------------------------------------------------------------------
// Segmented scan (groupby transforms: cumsum/cummin/cummax WITHIN key runs
// of a key-sorted column — the device form of pandas
// DataFrameGroupBy.cumsum, groupby.py "transform" family).
// State is the pair (f, v) — "range contains a segment head", "aggregate of
// the range's tail segment" — under
//   seg_comb((fa,va), (fb,vb)) = (fa|fb, fb ? vb : cs_comb(va,vb))
// which is associative but NOT commutative: every fold/scan below is an
// ordered left-to-right pass (the plain cumsum kernels' strided folds and
// shfl tree reductions are commutative-only and cannot be reused).
// NaN elements (f64) emit NaN at their position and contribute the identity
// to the running state (pandas cum* skipna); min/max propagate through the
// same cs_* helpers as hf_cumsum.  Three phases like hf_cumsum: per-tile
// summaries, one-block exclusive tile scan, per-element apply.
// ---------------------------------------------------------------------------
template <typename T, int OP>
__global__ void __launch_bounds__(BLOCK) k_seg_tiles(
    const T* __restrict__ in, const int64_t* __restrict__ heads, int64_t n,
    T* __restrict__ tile_v, unsigned* __restrict__ tile_f) {
  const int64_t t0 = (int64_t)blockIdx.x * FILT_TILE;
  const int64_t t1 = min(t0 + (int64_t)FILT_TILE, n);
  constexpr int PER = FILT_TILE / BLOCK;
  __shared__ T sv[FILT_TILE];
  __shared__ unsigned char sf[FILT_TILE];
  for (int j = 0; j < PER; ++j) {
    const int64_t i = t0 + (int64_t)j * BLOCK + threadIdx.x;
    if (i < t1) {
      sv[(int)(i - t0)] = in[i];
      sf[(int)(i - t0)] = heads[i] != 0;
    }
  }
  __syncthreads();
  const int s0 = threadIdx.x * PER;
  const int lim = (int)(t1 - t0);
  unsigned char f = 0;
  T v = cs_ident<T, OP>();
  for (int j = 0; j < PER; ++j) {
    if (s0 + j < lim) {
      const T x = sv[s0 + j];
      const T xv = (x != x) ? cs_ident<T, OP>() : x;
      if (sf[s0 + j]) { f = 1; v = xv; }
      else v = cs_comb<T, OP>(v, xv);
    }
  }
  __shared__ T bv[BLOCK];
  __shared__ unsigned char bf[BLOCK];
  bv[threadIdx.x] = v;
  bf[threadIdx.x] = f;
  __syncthreads();
  for (int off = 1; off < BLOCK; off <<= 1) {
    T pv = cs_ident<T, OP>();
    unsigned char pf = 0;
    if (threadIdx.x >= off) { pv = bv[threadIdx.x - off];
                              pf = bf[threadIdx.x - off]; }
    __syncthreads();
    const unsigned char nf = pf | bf[threadIdx.x];
    const T nv = bf[threadIdx.x] ? bv[threadIdx.x]
                                 : cs_comb<T, OP>(pv, bv[threadIdx.x]);
    __syncthreads();
    bv[threadIdx.x] = nv;
    bf[threadIdx.x] = nf;
    __syncthreads();
  }
  if (threadIdx.x == BLOCK - 1) {
    tile_v[blockIdx.x] = bv[threadIdx.x];
    tile_f[blockIdx.x] = bf[threadIdx.x];
  }
}

template <typename T, int OP>
__global__ void __launch_bounds__(1024) k_seg_scan_tiles(
    T* __restrict__ tile_v, unsigned* __restrict__ tile_f, int64_t ntiles) {
  __shared__ T carry_v;
  __shared__ unsigned carry_f;
  if (threadIdx.x == 0) { carry_v = cs_ident<T, OP>(); carry_f = 0; }
  __shared__ T bv[1024];
  __shared__ unsigned char bf[1024];
  __syncthreads();
  for (int64_t base = 0; base < ntiles; base += 1024) {
    const int64_t i = base + threadIdx.x;
    T v = cs_ident<T, OP>();
    unsigned char f = 0;
    if (i < ntiles) { v = tile_v[i]; f = (unsigned char)tile_f[i]; }
    bv[threadIdx.x] = v;
    bf[threadIdx.x] = f;
    __syncthreads();
    for (int off = 1; off < 1024; off <<= 1) {
      T pv = cs_ident<T, OP>();
      unsigned char pf = 0;
      if (threadIdx.x >= off) { pv = bv[threadIdx.x - off];
                                pf = bf[threadIdx.x - off]; }
      __syncthreads();
      const unsigned char nf = pf | bf[threadIdx.x];
      const T nv = bf[threadIdx.x] ? bv[threadIdx.x]
                                   : cs_comb<T, OP>(pv, bv[threadIdx.x]);
      __syncthreads();
      bv[threadIdx.x] = nv;
      bf[threadIdx.x] = nf;
      __syncthreads();
    }
    const T incl_v = bv[threadIdx.x];
    const unsigned char incl_f = bf[threadIdx.x];
    // exclusive = carry ⊕ previous lane's inclusive (identity at lane 0)
    const T pv = (threadIdx.x == 0) ? cs_ident<T, OP>()
                                    : bv[threadIdx.x - 1];
    const unsigned char pf = (threadIdx.x == 0) ? 0 : bf[threadIdx.x - 1];
    const unsigned ef = carry_f | pf;
    const T ev = pf ? pv : cs_comb<T, OP>(carry_v, pv);
    if (i < ntiles) { tile_v[i] = ev; tile_f[i] = ef; }
    __syncthreads();  // all lanes read carry before 1023 advances it
    if (threadIdx.x == 1023) {
      carry_v = incl_f ? incl_v : cs_comb<T, OP>(carry_v, incl_v);
      carry_f = carry_f | incl_f;
    }
    __syncthreads();
  }
}

template <typename T, int OP>
__global__ void __launch_bounds__(BLOCK) k_seg_apply(
    const T* __restrict__ in, const int64_t* __restrict__ heads, int64_t n,
    const T* __restrict__ tile_v, const unsigned* __restrict__ tile_f,
    T* __restrict__ out) {
  const int64_t t0 = (int64_t)blockIdx.x * FILT_TILE;
  const int64_t t1 = min(t0 + (int64_t)FILT_TILE, n);
  constexpr int PER = FILT_TILE / BLOCK;
  __shared__ T sv[FILT_TILE];
  __shared__ unsigned char sf[FILT_TILE];
  for (int j = 0; j < PER; ++j) {
    const int64_t i = t0 + (int64_t)j * BLOCK + threadIdx.x;
    if (i < t1) {
      sv[(int)(i - t0)] = in[i];
      sf[(int)(i - t0)] = heads[i] != 0;
    }
  }
  __syncthreads();
  const int s0 = threadIdx.x * PER;
  const int lim = (int)(t1 - t0);
  T lv[PER];
  unsigned char lf[PER];
  unsigned char f = 0;
  T v = cs_ident<T, OP>();
  for (int j = 0; j < PER; ++j) {
    if (s0 + j < lim) {
      const T x = sv[s0 + j];
      const T xv = (x != x) ? cs_ident<T, OP>() : x;
      if (sf[s0 + j]) { f = 1; v = xv; }
      else v = cs_comb<T, OP>(v, xv);
    }
    lv[j] = v;  // inclusive pair within the thread's run, per element
    lf[j] = f;
  }
  __shared__ T bv[BLOCK];
  __shared__ unsigned char bf[BLOCK];
  bv[threadIdx.x] = v;
  bf[threadIdx.x] = f;
  __syncthreads();
  for (int off = 1; off < BLOCK; off <<= 1) {
    T pv = cs_ident<T, OP>();
    unsigned char pf = 0;
    if (threadIdx.x >= off) { pv = bv[threadIdx.x - off];
                              pf = bf[threadIdx.x - off]; }
    __syncthreads();
    const unsigned char nf = pf | bf[threadIdx.x];
    const T nv = bf[threadIdx.x] ? bv[threadIdx.x]
                                 : cs_comb<T, OP>(pv, bv[threadIdx.x]);
    __syncthreads();
    bv[threadIdx.x] = nv;
    bf[threadIdx.x] = nf;
    __syncthreads();
  }
  // thread base = tile exclusive pair ⊕ preceding threads' inclusive pair
  const T pv = (threadIdx.x == 0) ? cs_ident<T, OP>() : bv[threadIdx.x - 1];
  const unsigned char pf = (threadIdx.x == 0) ? 0 : bf[threadIdx.x - 1];
  const T base_v = pf ? pv : cs_comb<T, OP>(tile_v[blockIdx.x], pv);
  __syncthreads();
  for (int j = 0; j < PER; ++j) {
    if (s0 + j < lim) {
      const T x = sv[s0 + j];
      const T r = lf[j] ? lv[j] : cs_comb<T, OP>(base_v, lv[j]);
      sv[s0 + j] = (x != x) ? x : r;
    }
  }
  __syncthreads();
  for (int j = 0; j < PER; ++j) {
    const int64_t i = t0 + (int64_t)j * BLOCK + threadIdx.x;
    if (i < t1) out[i] = sv[(int)(i - t0)];
  }
}

}  // namespace

struct hf_join {
  int64_t key_min, n_slots, n_right;
  int nr;
  unsigned long long* d_csr;   // [n_slots+1]
  unsigned* d_jidx;            // [n_right] right row index, key-grouped
  double* d_jval[GB_MAX_VALS]; // [nr][n_right] right values in CSR order
                               // (opaque 8 B payloads — i64 or f64)
  int dtypes[GB_MAX_VALS];     // per right column
};

// forward decls used by the map launch helpers (defined in the C ABI below)
extern "C" int hf_col_alloc(int64_t len, int dtype, hf_col** out);
extern "C" int hf_col_free(hf_col* col);
extern "C" int hf_map_scalar_i64(int op, const hf_col* in, int64_t scalar,
                                 hf_col** out);

namespace {
template <int OP>
int launch_map_f64(const hf_col* in, double s, hf_col* out) {
  const int64_t n = in->len;
  return timed_launch("map_f64", [&] {
    hipLaunchKernelGGL(k_map_f64<OP>, dim3(grid_for((n >> 1) + 1)), dim3(BLOCK), 0,
                       g.stream, (const double*)in->dptr, (double*)out->dptr, s, n);
  });
}
template <int OP>
int launch_map_i64(const hf_col* in, int64_t s, hf_col* out) {
  const int64_t n = in->len;
  return timed_launch("map_i64", [&] {
    hipLaunchKernelGGL(k_map_i64<OP>, dim3(grid_for((n >> 1) + 1)), dim3(BLOCK), 0,
                       g.stream, (const int64_t*)in->dptr, (int64_t*)out->dptr, s, n);
  });
}
}  // namespace

// ---------------------------------------------------------------------------
// C ABI
// ---------------------------------------------------------------------------

extern "C" {

int hf_init(int gpu) {
  if (g.inited && g.gpu == gpu) return HF_OK;
  if (g.inited) hf_shutdown();
  int n = 0;
  hipError_t e = hipGetDeviceCount(&n);
  if (e != hipSuccess) return set_hip_err("hf_init", e);
  if (gpu < 0 || gpu >= n)
    return set_err(HF_ERR_ARG, "hf_init", "gpu index out of range");
  HF_HIP("hf_init", hipSetDevice(gpu));
  HF_HIP("hf_init", hipStreamCreate(&g.stream));
  HF_HIP("hf_init", hipStreamCreate(&g.stream2));
  for (auto& ev : g.ov_ev)
    HF_HIP("hf_init", hipEventCreateWithFlags(&ev, hipEventDisableTiming));
  HF_HIP("hf_init", hipMalloc(&g.d_scratch, SCRATCH_BYTES));
  HF_HIP("hf_init", hipMemset(g.d_scratch, 0, SCRATCH_BYTES));
  g.gpu = gpu;
  g.inited = true;
  return HF_OK;
}

int hf_shutdown(void) {
  if (!g.inited) return HF_OK;
  hipStreamSynchronize(g.stream);
  if (g.stream2) hipStreamSynchronize(g.stream2);
  g_cache.trim();
  for (auto& p : g.pending) { hipEventDestroy(p.a); hipEventDestroy(p.b); }
  g.pending.clear();
  g.stats.clear();
  if (g.d_scratch) hipFree(g.d_scratch);
  for (auto& ev : g.ov_ev)
    if (ev) hipEventDestroy(ev);
  hipStreamDestroy(g.stream);
  if (g.stream2) hipStreamDestroy(g.stream2);
  g = State{};
  return HF_OK;
}

int hf_device_count(int* out) {
  int n = 0;
  hipError_t e = hipGetDeviceCount(&n);
  if (e != hipSuccess) { *out = 0; return set_hip_err("hf_device_count", e); }
  *out = n;
  return HF_OK;
}

const char* hf_last_error(void) { return g_err.c_str(); }

int hf_sync(void) {
  HF_NEED_INIT("hf_sync");
  HF_HIP("hf_sync", hipStreamSynchronize(g.stream));
  return HF_OK;
}

// ---- memory ----

int hf_col_alloc(int64_t len, int dtype, hf_col** out) {
  HF_NEED_INIT("hf_col_alloc");
  if (len < 0 || dtype_size(dtype) == 0 || !out)
    return set_err(HF_ERR_ARG, "hf_col_alloc", "bad len/dtype");
  void* d = nullptr;
  int64_t bytes = len * dtype_size(dtype);
  if (bytes == 0) bytes = 8;  // keep zero-length columns addressable
  HF_HIP("hf_col_alloc", dev_alloc((void**)&d, bytes, g.stream));
  hf_col* c = new hf_col{d, len, dtype, g.gpu};
  *out = c;
  return HF_OK;
}

int hf_put(const void* host, int64_t len, int dtype, hf_col** out) {
  HF_NEED_INIT("hf_put");
  int rc = hf_col_alloc(len, dtype, out);
  if (rc != HF_OK) return rc;
  if (len > 0)
    HF_HIP("hf_put", hipMemcpyAsync((*out)->dptr, host, len * dtype_size(dtype),
                                    hipMemcpyHostToDevice, g.stream));
  return HF_OK;
}

int hf_get(const hf_col* col, void* host) {
  HF_NEED_INIT("hf_get");
  if (!col || !host) return set_err(HF_ERR_ARG, "hf_get", "null");
  if (col->len > 0)
    HF_HIP("hf_get", hipMemcpyAsync(host, col->dptr, col->len * dtype_size(col->dtype),
                                    hipMemcpyDeviceToHost, g.stream));
  HF_HIP("hf_get", hipStreamSynchronize(g.stream));
  return HF_OK;
}

int hf_col_free(hf_col* col) {
  if (!col) return HF_OK;
  if (g.inited && col->dptr) dev_free(col->dptr, g.stream);
  if (col->d_hist) free(col->d_hist);  // host-side cached histogram
  if (g.inited && col->d_k32) dev_free(col->d_k32, g.stream);
  if (g.inited && col->d_curinit) dev_free(col->d_curinit, g.stream);
  if (g.inited && col->d_work) dev_free(col->d_work, g.stream);
  delete col;
  return HF_OK;
}

int64_t hf_col_len(const hf_col* c) { return c ? c->len : -1; }
int hf_col_dtype(const hf_col* c) { return c ? c->dtype : -1; }
uintptr_t hf_col_dptr(const hf_col* c) { return c ? (uintptr_t)c->dptr : 0; }

int hf_alloc_raw(int64_t bytes, uintptr_t* dptr) {
  HF_NEED_INIT("hf_alloc_raw");
  void* d = nullptr;
  HF_HIP("hf_alloc_raw", dev_alloc((void**)&d, bytes, g.stream));
  *dptr = (uintptr_t)d;
  return HF_OK;
}
int hf_free_raw(uintptr_t dptr) {
  HF_NEED_INIT("hf_free_raw");
  HF_HIP("hf_free_raw", dev_free((void*)dptr, g.stream));
  return HF_OK;
}
int hf_memset_raw(uintptr_t dptr, int value, int64_t bytes) {
  HF_NEED_INIT("hf_memset_raw");
  HF_HIP("hf_memset_raw", hipMemsetAsync((void*)dptr, value, bytes, g.stream));
  return HF_OK;
}

// ---- map ----

int hf_map_scalar(int op, const hf_col* in, double scalar, hf_col** out) {
  HF_NEED_INIT("hf_map_scalar");
  if (!in || !out) return set_err(HF_ERR_ARG, "hf_map_scalar", "null");
  if (op == HF_MAP_CAST_F64) {
    if (in->dtype != HF_INT64)
      return set_err(HF_ERR_ARG, "hf_map_scalar", "cast_f64 needs int64 input");
    int rc = hf_col_alloc(in->len, HF_FLOAT64, out);
    if (rc != HF_OK) return rc;
    return timed_launch("cast_i64_f64", [&] {
      hipLaunchKernelGGL(k_cast_i64_f64, dim3(grid_for(in->len)), dim3(BLOCK), 0,
                         g.stream, (const int64_t*)in->dptr, (double*)(*out)->dptr,
                         in->len);
    });
  }
  if (op == HF_MAP_CAST_I64) {
    if (in->dtype != HF_FLOAT64)
      return set_err(HF_ERR_ARG, "hf_map_scalar", "cast_i64 needs f64 input");
    int rc = hf_col_alloc(in->len, HF_INT64, out);
    if (rc != HF_OK) return rc;
    return timed_launch("cast_f64_i64", [&] {
      hipLaunchKernelGGL(k_cast_f64_i64, dim3(grid_for(in->len)), dim3(BLOCK), 0,
                         g.stream, (const double*)in->dptr,
                         (int64_t*)(*out)->dptr, in->len);
    });
  }
  if (in->dtype == HF_INT64) return hf_map_scalar_i64(op, in, (int64_t)scalar, out);
  int rc = hf_col_alloc(in->len, HF_FLOAT64, out);
  if (rc != HF_OK) return rc;
  switch (op) {
    case HF_MAP_ADD:    rc = launch_map_f64<HF_MAP_ADD>(in, scalar, *out); break;
    case HF_MAP_SUB:    rc = launch_map_f64<HF_MAP_SUB>(in, scalar, *out); break;
    case HF_MAP_RSUB:   rc = launch_map_f64<HF_MAP_RSUB>(in, scalar, *out); break;
    case HF_MAP_MUL:    rc = launch_map_f64<HF_MAP_MUL>(in, scalar, *out); break;
    case HF_MAP_DIV:    rc = launch_map_f64<HF_MAP_DIV>(in, scalar, *out); break;
    case HF_MAP_RDIV:   rc = launch_map_f64<HF_MAP_RDIV>(in, scalar, *out); break;
    case HF_MAP_FILLNA: rc = launch_map_f64<HF_MAP_FILLNA>(in, scalar, *out); break;
    case HF_MAP_ABS:    rc = launch_map_f64<HF_MAP_ABS>(in, scalar, *out); break;
    case HF_MAP_NEG:    rc = launch_map_f64<HF_MAP_NEG>(in, scalar, *out); break;
    case HF_MAP_SQRT:   rc = launch_map_f64<HF_MAP_SQRT>(in, scalar, *out); break;
    case HF_MAP_MIN:    rc = launch_map_f64<HF_MAP_MIN>(in, scalar, *out); break;
    case HF_MAP_MAX:    rc = launch_map_f64<HF_MAP_MAX>(in, scalar, *out); break;
    case HF_MAP_ROUND:  rc = launch_map_f64<HF_MAP_ROUND>(in, scalar, *out); break;
    default:
      hf_col_free(*out);
      *out = nullptr;
      return set_err(HF_ERR_UNSUPPORTED, "hf_map_scalar", "unknown op");
  }
  if (rc != HF_OK) { hf_col_free(*out); *out = nullptr; }
  return rc;
}

int hf_map_scalar_i64(int op, const hf_col* in, int64_t scalar, hf_col** out) {
  HF_NEED_INIT("hf_map_scalar_i64");
  if (!in || !out) return set_err(HF_ERR_ARG, "hf_map_scalar_i64", "null");
  if (in->dtype != HF_INT64)
    return set_err(HF_ERR_ARG, "hf_map_scalar_i64", "int64 column required");
  int rc = hf_col_alloc(in->len, HF_INT64, out);
  if (rc != HF_OK) return rc;
  switch (op) {
    case HF_MAP_ADD:  rc = launch_map_i64<HF_MAP_ADD>(in, scalar, *out); break;
    case HF_MAP_SUB:  rc = launch_map_i64<HF_MAP_SUB>(in, scalar, *out); break;
    case HF_MAP_RSUB: rc = launch_map_i64<HF_MAP_RSUB>(in, scalar, *out); break;
    case HF_MAP_MUL:  rc = launch_map_i64<HF_MAP_MUL>(in, scalar, *out); break;
    case HF_MAP_ABS:  rc = launch_map_i64<HF_MAP_ABS>(in, scalar, *out); break;
    case HF_MAP_NEG:  rc = launch_map_i64<HF_MAP_NEG>(in, scalar, *out); break;
    case HF_MAP_MIN:  rc = launch_map_i64<HF_MAP_MIN>(in, scalar, *out); break;
    case HF_MAP_MAX:  rc = launch_map_i64<HF_MAP_MAX>(in, scalar, *out); break;
    case HF_MAP_IDIV:
      if (scalar == 0) {
        hf_col_free(*out);
        *out = nullptr;
        return set_err(HF_ERR_ARG, "hf_map_scalar_i64", "floordiv by zero");
      }
      rc = launch_map_i64<HF_MAP_IDIV>(in, scalar, *out);
      break;
    case HF_MAP_IMOD:
      if (scalar == 0) {
        hf_col_free(*out);
        *out = nullptr;
        return set_err(HF_ERR_ARG, "hf_map_scalar_i64", "mod by zero");
      }
      rc = launch_map_i64<HF_MAP_IMOD>(in, scalar, *out);
      break;
    default:
      hf_col_free(*out);
      *out = nullptr;
      return set_err(HF_ERR_UNSUPPORTED, "hf_map_scalar_i64",
                     "op not defined for int64 (div promotes via cast_f64)");
  }
  if (rc != HF_OK) { hf_col_free(*out); *out = nullptr; }
  return rc;
}

// ---- binary ----

int hf_binary(int op, const hf_col* a, const hf_col* b, hf_col** out) {
  HF_NEED_INIT("hf_binary");
  if (!a || !b || !out) return set_err(HF_ERR_ARG, "hf_binary", "null");
  if (a->len != b->len)
    return set_err(HF_ERR_ARG, "hf_binary", "length mismatch");
  if (a->dtype != b->dtype)
    return set_err(HF_ERR_ARG, "hf_binary", "dtype mismatch (cast first)");
  const bool f64 = a->dtype == HF_FLOAT64;
  if (!f64 && op == HF_BIN_DIV)
    return set_err(HF_ERR_UNSUPPORTED, "hf_binary", "int64 div promotes via cast_f64");
  int rc = hf_col_alloc(a->len, a->dtype, out);
  if (rc != HF_OK) return rc;
  const int64_t n = a->len;
  // dispatch with typed wrappers
  auto Lf = [&](auto opTag) {
    constexpr int O = decltype(opTag)::value;
    return timed_launch("bin_f64", [&] {
      hipLaunchKernelGGL((k_bin<O, double, double2>), dim3(grid_for((n >> 1) + 1)),
                         dim3(BLOCK), 0, g.stream, (const double*)a->dptr,
                         (const double*)b->dptr, (double*)(*out)->dptr, n);
    });
  };
  auto Li = [&](auto opTag) {
    constexpr int O = decltype(opTag)::value;
    return timed_launch("bin_i64", [&] {
      hipLaunchKernelGGL((k_bin<O, int64_t, longlong2>), dim3(grid_for((n >> 1) + 1)),
                         dim3(BLOCK), 0, g.stream, (const int64_t*)a->dptr,
                         (const int64_t*)b->dptr, (int64_t*)(*out)->dptr, n);
    });
  };
  switch (op) {
    case HF_BIN_ADD: rc = f64 ? Lf(std::integral_constant<int, HF_BIN_ADD>{})
                              : Li(std::integral_constant<int, HF_BIN_ADD>{}); break;
    case HF_BIN_SUB: rc = f64 ? Lf(std::integral_constant<int, HF_BIN_SUB>{})
                              : Li(std::integral_constant<int, HF_BIN_SUB>{}); break;
    case HF_BIN_MUL: rc = f64 ? Lf(std::integral_constant<int, HF_BIN_MUL>{})
                              : Li(std::integral_constant<int, HF_BIN_MUL>{}); break;
    case HF_BIN_DIV: rc = Lf(std::integral_constant<int, HF_BIN_DIV>{}); break;
    case HF_BIN_MIN: rc = f64 ? Lf(std::integral_constant<int, HF_BIN_MIN>{})
                              : Li(std::integral_constant<int, HF_BIN_MIN>{}); break;
    case HF_BIN_MAX: rc = f64 ? Lf(std::integral_constant<int, HF_BIN_MAX>{})
                              : Li(std::integral_constant<int, HF_BIN_MAX>{}); break;
    default:
      hf_col_free(*out);
      *out = nullptr;
      return set_err(HF_ERR_UNSUPPORTED, "hf_binary", "unknown op");
  }
  if (rc != HF_OK) { hf_col_free(*out); *out = nullptr; }
  return rc;
}

// ---- reduce ----

int hf_reduce(const hf_col* in, hf_reduce_result* out) {
  HF_NEED_INIT("hf_reduce");
  if (!in || !out) return set_err(HF_ERR_ARG, "hf_reduce", "null");
  ReduceAccF64* dF = (ReduceAccF64*)g.d_scratch;
  ReduceAccI64* dI = (ReduceAccI64*)((char*)g.d_scratch + 64);
  hipLaunchKernelGGL(k_reduce_init, dim3(1), dim3(1), 0, g.stream, dF, dI);
  const int64_t n = in->len;
  int rc;
  if (in->dtype == HF_FLOAT64) {
    rc = timed_launch("reduce_f64", [&] {
      hipLaunchKernelGGL(k_reduce_f64, dim3(grid_for((n >> 1) + 1)), dim3(BLOCK), 0,
                         g.stream, (const double*)in->dptr, n, dF);
    });
  } else {
    rc = timed_launch("reduce_i64", [&] {
      hipLaunchKernelGGL(k_reduce_i64, dim3(grid_for((n >> 1) + 1)), dim3(BLOCK), 0,
                         g.stream, (const int64_t*)in->dptr, n, dI);
    });
  }
  if (rc != HF_OK) return rc;
  ReduceAccF64 hF;
  ReduceAccI64 hI;
  HF_HIP("hf_reduce", hipMemcpyAsync(&hF, dF, sizeof(hF), hipMemcpyDeviceToHost, g.stream));
  HF_HIP("hf_reduce", hipMemcpyAsync(&hI, dI, sizeof(hI), hipMemcpyDeviceToHost, g.stream));
  HF_HIP("hf_reduce", hipStreamSynchronize(g.stream));
  if (in->dtype == HF_FLOAT64) {
    out->sum = hF.sum;
    out->count = (int64_t)hF.count;
    out->mn = hF.mn;
    out->mx = hF.mx;
    out->isum = (int64_t)hF.sum;
    out->imn = (int64_t)hF.mn;
    out->imx = (int64_t)hF.mx;
  } else {
    out->sum = (double)hI.sum;
    out->count = (int64_t)hI.count;
    out->mn = (double)hI.mn;
    out->mx = (double)hI.mx;
    out->isum = hI.sum;
    out->imn = hI.mn;
    out->imx = hI.mx;
  }
  return HF_OK;
}

// ---- groupby ----

}  // extern "C" (the radix-path helpers below are C++ templates)

namespace {

// Per-bucket histogram of an (immutable) key column, cached host-side on the
// hf_col — repeated groupbys on the same keys skip the P0 pass entirely.
int ensure_host_hist(hf_col* keys, int64_t key_min, int64_t n_slots,
                     int64_t nb, int range_log) {
  if (keys->d_hist && keys->hist_kmin == key_min && keys->hist_nb == nb)
    return HF_OK;
  if (keys->d_hist) { free(keys->d_hist); keys->d_hist = nullptr; }
  // the radix layout cache derives from this histogram: invalidate with it
  if (keys->d_curinit) {
    dev_free(keys->d_curinit, g.stream);
    keys->d_curinit = nullptr;
  }
  if (keys->d_work) {
    dev_free(keys->d_work, g.stream);
    keys->d_work = nullptr;
  }
  keys->radix_rl = 0;
  unsigned long long* d_h = nullptr;
  HF_HIP("gb_hist", dev_alloc((void**)&d_h, nb * 8, g.stream));
  HF_HIP("gb_hist", hipMemsetAsync(d_h, 0, nb * 8, g.stream));
  const int64_t n = keys->len;
  int rc = timed_launch("gb_hist", [&] {
    hipLaunchKernelGGL(k_gb_hist, dim3((uint32_t)grid_for(n)), dim3(BLOCK),
                       (uint32_t)(nb * 4), g.stream, (const int64_t*)keys->dptr,
                       n, key_min, n_slots, (int)nb, range_log, d_h);
  });
  if (rc != HF_OK) return rc;
  int64_t* h = (int64_t*)malloc(nb * 8);
  HF_HIP("gb_hist", hipMemcpyAsync(h, d_h, nb * 8, hipMemcpyDeviceToHost,
                                   g.stream));
  HF_HIP("gb_hist", hipStreamSynchronize(g.stream));
  HF_HIP("gb_hist", dev_free(d_h, g.stream));
  keys->d_hist = h;
  keys->hist_kmin = key_min;
  keys->hist_nb = nb;
  return HF_OK;
}

template <int AOP>
int gb_dense_path(const hf_col* keys, const GbPtrs& ptrs, int nvals,
                  int64_t key_min, int64_t n_slots, uintptr_t sums,
                  uintptr_t rowcnt, uintptr_t counts,
                  unsigned long long* d_err) {
  const int64_t n = keys->len;
  const bool cnt = counts != 0;
  int64_t grid = n / 65536 + 1;
  if (grid > 512) grid = 512;
  auto L = [&](auto rTag, auto cTag, auto vTag, const double* v, double* gs,
               unsigned long long* gc) {
    constexpr bool R = decltype(rTag)::value, C = decltype(cTag)::value,
                   V = decltype(vTag)::value;
    const uint32_t lds =
        (uint32_t)(n_slots * ((V ? 8 : 0) + 1 + (C ? 4 : 0)));
    return timed_launch("gb_dense", [&] {
      hipLaunchKernelGGL((k_gb_dense<R, C, V, AOP>), dim3((uint32_t)grid),
                         dim3(512), lds, g.stream, (const int64_t*)keys->dptr,
                         v, n, key_min, n_slots, gs,
                         (unsigned long long*)rowcnt, gc, d_err);
    });
  };
  using T = std::true_type;
  using F = std::false_type;
  if (nvals == 0) return L(T{}, F{}, F{}, nullptr, nullptr, nullptr);
  for (int c = 0; c < nvals; ++c) {
    double* gs = (double*)sums + (int64_t)c * n_slots;
    unsigned long long* gc =
        cnt ? (unsigned long long*)counts + (int64_t)c * n_slots : nullptr;
    int rc;
    if (c == 0)
      rc = cnt ? L(T{}, T{}, T{}, ptrs.vals[c], gs, gc)
               : L(T{}, F{}, T{}, ptrs.vals[c], gs, gc);
    else
      rc = cnt ? L(F{}, T{}, T{}, ptrs.vals[c], gs, gc)
               : L(F{}, F{}, T{}, ptrs.vals[c], gs, gc);
    if (rc != HF_OK) return rc;
  }
  return HF_OK;
}

int ensure_keys32(hf_col* keys, int64_t key_min) {
  if (keys->d_k32 && keys->k32_min == key_min) return HF_OK;
  if (keys->d_k32) { dev_free(keys->d_k32, g.stream); keys->d_k32 = nullptr; }
  const int64_t n = keys->len;
  HF_HIP("keys32", dev_alloc(&keys->d_k32, (n > 0 ? n : 1) * 4, g.stream));
  int rc = timed_launch("gb_keys32", [&] {
    hipLaunchKernelGGL(k_conv_keys32, dim3((uint32_t)grid_for(n)), dim3(BLOCK),
                       0, g.stream, (const int64_t*)keys->dptr, n, key_min,
                       (unsigned*)keys->d_k32);
  });
  if (rc != HF_OK) return rc;
  keys->k32_min = key_min;
  return HF_OK;
}

template <int RL, int AOP>
int gb_radix_path(hf_col* keys, const GbPtrs& ptrs, int nvals, int64_t key_min,
                  int64_t n_slots, uintptr_t sums, uintptr_t rowcnt,
                  uintptr_t counts, unsigned long long* d_err,
                  bool with_rowcnt = true) {
  const int64_t n = keys->len;
  const int64_t nb = (n_slots + (1 << RL) - 1) >> RL;
  int rc = ensure_host_hist(keys, key_min, n_slots, nb, RL);
  if (rc != HF_OK) return rc;
  rc = ensure_keys32(keys, key_min);
  if (rc != HF_OK) return rc;
  const int64_t* h = keys->d_hist ? (const int64_t*)keys->d_hist : nullptr;
  // exact per-bucket regions, 64-row aligned; u32 cursors cap one partition
  // at ~4.29e9 rows (shard larger frames).  The cursor-init array and the
  // work items derive only from the cached histogram, so they cache on the
  // immutable key column too: steady-state groupby calls copy cursors D2D
  // and never touch the host (no pageable H2D, no mid-op sync).
  if (!keys->d_curinit || keys->radix_rl != RL) {
    if (keys->d_curinit) {
      dev_free(keys->d_curinit, g.stream);
      keys->d_curinit = nullptr;
    }
    if (keys->d_work) {
      dev_free(keys->d_work, g.stream);
      keys->d_work = nullptr;
    }
    std::vector<unsigned> cur_init((size_t)nb);
    std::vector<GbWorkItem> work;
    int64_t off = 0;
    for (int64_t b = 0; b < nb; ++b) {
      cur_init[b] = (unsigned)off;
      const int64_t rows_b = h[b];
      int64_t done = 0;
      while (done < rows_b) {
        const int64_t len = std::min(AGG_CHUNK, rows_b - done);
        work.push_back(GbWorkItem{off + done, (int32_t)b, (int32_t)len});
        done += len;
      }
      off += (rows_b + 63) & ~63LL;
    }
    if (off > 0xFFFFFFF0LL)
      return set_err(HF_ERR_UNSUPPORTED, "hf_groupby_accum",
                     "partition too large for radix scatter (shard it)");
    HF_HIP("gb_radix", dev_alloc(&keys->d_curinit, nb * 4, g.stream));
    HF_HIP("gb_radix",
           dev_alloc(&keys->d_work,
                     std::max<size_t>(work.size(), 1) * sizeof(GbWorkItem),
                     g.stream));
    HF_HIP("gb_radix", hipMemcpyAsync(keys->d_curinit, cur_init.data(),
                                      nb * 4, hipMemcpyHostToDevice,
                                      g.stream));
    if (!work.empty())
      HF_HIP("gb_radix",
             hipMemcpyAsync(keys->d_work, work.data(),
                            work.size() * sizeof(GbWorkItem),
                            hipMemcpyHostToDevice, g.stream));
    // host vectors must outlive the async H2D of pageable memory
    HF_HIP("gb_radix", hipStreamSynchronize(g.stream));
    keys->work_n = (int64_t)work.size();
    keys->radix_rows = off;
    keys->radix_rl = RL;
  }
  const int64_t off = keys->radix_rows;
  GbWorkItem* d_work = (GbWorkItem*)keys->d_work;
  double* r0 = nullptr;
  double* r1 = nullptr;
  unsigned short* rk = nullptr;
  unsigned* d_cur = nullptr;
  const int64_t alloc_rows = off > 0 ? off : 64;
  if (nvals > 0)
    HF_HIP("gb_radix", dev_alloc((void**)&r0, alloc_rows * 8, g.stream));
  if (nvals > 1)
    HF_HIP("gb_radix", dev_alloc((void**)&r1, alloc_rows * 8, g.stream));
  HF_HIP("gb_radix", dev_alloc((void**)&rk, alloc_rows * 2, g.stream));
  HF_HIP("gb_radix", dev_alloc((void**)&d_cur, nb * 4, g.stream));
  HF_HIP("gb_radix", hipMemcpyAsync(d_cur, keys->d_curinit, nb * 4,
                                    hipMemcpyDeviceToDevice, g.stream));

  // ---- overlapped path: scatter in 4 chunks on the main stream; after
  // each chunk the consumer stream aggregates the region slices that
  // chunk completed (device-built work items from cursor snapshots).
  // MEASURED NEGATIVE (round 2, profiles/r02): both kernels want >72 KB
  // LDS, so concurrent scheduling PARTITIONS the CUs between them and
  // each phase runs ~1.5x slower (uniform 9.2 ms vs 7.0 serial) — the
  // overlap never recovers the loss.  Kept behind HF_GB_OVERLAP=1 for
  // future experiments; default is the serial two-phase path.
  static const bool overlap_env = [] {
    const char* e = getenv("HF_GB_OVERLAP");
    return e && e[0] == '1';
  }();
  const bool overlap =
      overlap_env && n >= (64LL << 20) && nvals <= 2 && nb <= 1024;
  if (overlap) {
    const int NCH = 4;
    const double fracs[NCH] = {0.30, 0.30, 0.30, 0.10};
    unsigned* d_snap[NCH];
    for (int c = 0; c < NCH; ++c)
      HF_HIP("gb_radix", dev_alloc((void**)&d_snap[c], nb * 4, g.stream));
    int64_t bounds[NCH + 1];
    bounds[0] = 0;
    double accf = 0;
    for (int c = 0; c < NCH - 1; ++c) {
      accf += fracs[c];
      bounds[c + 1] = ((int64_t)(n * accf)) & ~1LL;
    }
    bounds[NCH] = n;
    auto scat_chunk = [&](auto nvTag, auto rptTag, auto blkTag, int c) {
      constexpr int NVv = decltype(nvTag)::value;
      constexpr int RPTv = decltype(rptTag)::value;
      constexpr int BLKv = decltype(blkTag)::value;
      const int64_t tile_sz = (int64_t)BLKv * RPTv;
      const int64_t rows_c = bounds[c + 1] - bounds[c];
      const int64_t ntiles = (rows_c + tile_sz - 1) / tile_sz;
      const uint32_t sgrid =
          (uint32_t)std::min<int64_t>(std::max<int64_t>(ntiles, 1), 2048);
      const uint32_t lds =
          (uint32_t)(tile_sz * (8 * NVv + 4) + nb * 12 + 16);
      return timed_launch("gb_scatter", [&] {
        hipLaunchKernelGGL((k_gb_scatter<NVv, RPTv, BLKv, RL, true>),
                           dim3(sgrid), dim3(BLKv), lds, g.stream,
                           (const int64_t*)keys->dptr,
                           (const unsigned*)keys->d_k32, ptrs.vals[0],
                           ptrs.vals[1], bounds[c], bounds[c + 1], key_min,
                           n_slots, (int)nb, d_cur, r0, r1, rk, d_err);
      });
    };
    // per-chunk device work items (k_gb_make_work) + the regular
    // atomic-merge P2 with an upper-bound grid gated by the device count
    GbWorkItem* d_workc[NCH];
    int* d_nwork[NCH];
    uint32_t ub[NCH];
    for (int c = 0; c < NCH; ++c) {
      const int64_t rows_c = bounds[c + 1] - bounds[c];
      ub[c] = (uint32_t)(nb + rows_c / AGG_CHUNK + 1);
      HF_HIP("gb_radix",
             dev_alloc((void**)&d_workc[c], ub[c] * sizeof(GbWorkItem),
                       g.stream));
      HF_HIP("gb_radix", dev_alloc((void**)&d_nwork[c], 4, g.stream));
    }
    auto agg_chunk = [&](auto rTag, auto cTag, auto vTag, const double* v,
                         double* gs, unsigned long long* gc, int c) {
      constexpr bool R = decltype(rTag)::value, C = decltype(cTag)::value,
                     V = decltype(vTag)::value;
      return timed_launch_on("gb_bucket_agg", g.stream2, [&] {
        hipLaunchKernelGGL((k_gb_bucket_agg<R, C, V, RL, AOP>),
                           dim3(ub[c]), dim3(512), 0, g.stream2, v, rk,
                           d_workc[c], d_nwork[c], n_slots, gs,
                           (unsigned long long*)rowcnt, gc);
      });
    };
    using T = std::true_type;
    using F = std::false_type;
    const bool cnt = counts != 0;
    for (int c = 0; c < NCH && rc == HF_OK; ++c) {
      rc = nvals == 0 ? scat_chunk(std::integral_constant<int, 0>{},
                                   std::integral_constant<int, 12>{},
                                   std::integral_constant<int, 1024>{}, c)
           : nvals == 1 ? scat_chunk(std::integral_constant<int, 1>{},
                                     std::integral_constant<int, 12>{},
                                     std::integral_constant<int, 1024>{}, c)
                        : scat_chunk(std::integral_constant<int, 2>{},
                                     std::integral_constant<int, 12>{},
                                     std::integral_constant<int, 512>{}, c);
      if (rc != HF_OK) break;
      HF_HIP("gb_radix", hipMemcpyAsync(d_snap[c], d_cur, nb * 4,
                                        hipMemcpyDeviceToDevice, g.stream));
      HF_HIP("gb_radix", hipEventRecord(g.ov_ev[c], g.stream));
      HF_HIP("gb_radix", hipStreamWaitEvent(g.stream2, g.ov_ev[c], 0));
      const unsigned* clo =
          c == 0 ? (const unsigned*)keys->d_curinit : d_snap[c - 1];
      const unsigned* chi = d_snap[c];
      rc = timed_launch_on("gb_make_work", g.stream2, [&] {
        hipLaunchKernelGGL(k_gb_make_work, dim3(1), dim3(256), 0, g.stream2,
                           clo, chi, (int)nb, (int)AGG_CHUNK, d_workc[c],
                           d_nwork[c]);
      });
      if (rc != HF_OK) break;
      if (nvals == 0) {
        rc = agg_chunk(T{}, F{}, F{}, nullptr, nullptr, nullptr, c);
      } else {
        for (int col = 0; col < nvals && rc == HF_OK; ++col) {
          const double* v = col == 0 ? r0 : r1;
          double* gs = (double*)sums + (int64_t)col * n_slots;
          unsigned long long* gc =
              cnt ? (unsigned long long*)counts + (int64_t)col * n_slots
                  : nullptr;
          if (col == 0 && with_rowcnt)
            rc = cnt ? agg_chunk(T{}, T{}, T{}, v, gs, gc, c)
                     : agg_chunk(T{}, F{}, T{}, v, gs, gc, c)
            ;
          else
            rc = cnt ? agg_chunk(F{}, T{}, T{}, v, gs, gc, c)
                     : agg_chunk(F{}, F{}, T{}, v, gs, gc, c);
        }
      }
    }
    // fence the consumer stream back onto the main stream BEFORE the
    // payload buffers are released (keeps the allocator's single-stream
    // reuse argument intact)
    HF_HIP("gb_radix", hipEventRecord(g.ov_ev[NCH], g.stream2));
    HF_HIP("gb_radix", hipStreamWaitEvent(g.stream, g.ov_ev[NCH], 0));
    for (int c = 0; c < NCH; ++c) {
      dev_free(d_snap[c], g.stream);
      dev_free(d_workc[c], g.stream);
      dev_free(d_nwork[c], g.stream);
    }
    if (r0) dev_free(r0, g.stream);
    if (r1) dev_free(r1, g.stream);
    dev_free(rk, g.stream);
    dev_free(d_cur, g.stream);
    return rc;
  }
  // P1 scatter (pipelined): 12288-row tiles as 1024x12 for <=1 value column
  // — 147 KB LDS, 1 block/CU of 16 waves (round-2 probe: 1024x12 edges out
  // 512x24 and both beat the round-1 kernel by ~17%); 6144-row tiles as
  // 512x12 for 2 columns
  auto scat = [&](auto nvTag, auto rptTag, auto blkTag) {
    constexpr int NVv = decltype(nvTag)::value;
    constexpr int RPTv = decltype(rptTag)::value;
    constexpr int BLKv = decltype(blkTag)::value;
    const int64_t tile_sz = (int64_t)BLKv * RPTv;
    const int64_t ntiles = ((n & ~1LL) + tile_sz - 1) / tile_sz;
    const uint32_t sgrid =
        (uint32_t)std::min<int64_t>(std::max<int64_t>(ntiles, 1), 2048);
    const uint32_t lds =
        (uint32_t)(tile_sz * (8 * NVv + 4) + nb * 12 + 16);
    return timed_launch("gb_scatter", [&] {
      hipLaunchKernelGGL((k_gb_scatter<NVv, RPTv, BLKv, RL, true>),
                         dim3(sgrid), dim3(BLKv), lds, g.stream,
                         (const int64_t*)keys->dptr,
                         (const unsigned*)keys->d_k32, ptrs.vals[0],
                         ptrs.vals[1], (int64_t)0, n, key_min, n_slots,
                         (int)nb, d_cur, r0, r1, rk, d_err);
    });
  };
  rc = nvals == 0 ? scat(std::integral_constant<int, 0>{},
                         std::integral_constant<int, 12>{},
                         std::integral_constant<int, 1024>{})
       : nvals == 1 ? scat(std::integral_constant<int, 1>{},
                           std::integral_constant<int, 12>{},
                           std::integral_constant<int, 1024>{})
                    : scat(std::integral_constant<int, 2>{},
                           std::integral_constant<int, 12>{},
                           std::integral_constant<int, 512>{});
  if (rc != HF_OK) return rc;
  // P2 aggregate, one column per launch
  const bool cnt = counts != 0;
  const uint32_t agrid = (uint32_t)keys->work_n;
  auto agg = [&](auto rTag, auto cTag, auto vTag, const double* v, double* gs,
                 unsigned long long* gc) {
    constexpr bool R = decltype(rTag)::value, C = decltype(cTag)::value,
                   V = decltype(vTag)::value;
    return timed_launch("gb_bucket_agg", [&] {
      hipLaunchKernelGGL((k_gb_bucket_agg<R, C, V, RL, AOP>), dim3(agrid),
                         dim3(512), 0, g.stream, v, rk, d_work, nullptr,
                         n_slots, gs, (unsigned long long*)rowcnt, gc);
    });
  };
  using T = std::true_type;
  using F = std::false_type;
  if (agrid > 0) {
    if (nvals == 0) {
      rc = agg(T{}, F{}, F{}, nullptr, nullptr, nullptr);
    } else {
      for (int c = 0; c < nvals && rc == HF_OK; ++c) {
        const double* v = c == 0 ? r0 : r1;
        double* gs = (double*)sums + (int64_t)c * n_slots;
        unsigned long long* gc =
            cnt ? (unsigned long long*)counts + (int64_t)c * n_slots : nullptr;
        if (c == 0 && with_rowcnt)
          rc = cnt ? agg(T{}, T{}, T{}, v, gs, gc)
                   : agg(T{}, F{}, T{}, v, gs, gc);
        else
          rc = cnt ? agg(F{}, T{}, T{}, v, gs, gc)
                   : agg(F{}, F{}, T{}, v, gs, gc);
      }
    }
  }
  if (r0) dev_free(r0, g.stream);
  if (r1) dev_free(r1, g.stream);
  dev_free(rk, g.stream);
  dev_free(d_cur, g.stream);
  return rc;
}

}  // namespace

extern "C" {

int hf_groupby_accum(const hf_col* keys, const hf_col* const* vals, int nvals,
                     int agg_op, int64_t key_min, int64_t n_slots,
                     uintptr_t sums, uintptr_t rowcnt, uintptr_t counts) {
  HF_NEED_INIT("hf_groupby_accum");
  if (!keys || !vals || nvals < 0 || nvals > GB_MAX_VALS)
    return set_err(HF_ERR_ARG, "hf_groupby_accum", "bad args (nvals<=8)");
  if (agg_op < HF_AGG_SUM || agg_op > HF_AGG_MAX)
    return set_err(HF_ERR_ARG, "hf_groupby_accum", "bad agg_op");
  if (keys->dtype != HF_INT64)
    return set_err(HF_ERR_ARG, "hf_groupby_accum", "keys must be int64");
  if (n_slots <= 0) return set_err(HF_ERR_ARG, "hf_groupby_accum", "n_slots<=0");
  GbPtrs ptrs{};
  for (int c = 0; c < nvals; ++c) {
    if (!vals[c] || vals[c]->dtype != HF_FLOAT64 || vals[c]->len != keys->len)
      return set_err(HF_ERR_ARG, "hf_groupby_accum",
                     "vals must be float64 columns of keys' length");
    ptrs.vals[c] = (const double*)vals[c]->dptr;
  }
  const int64_t n = keys->len;
  unsigned long long* d_err =
      (unsigned long long*)((char*)g.d_scratch + SCRATCH_GB_ERR);
  // path selection (DESIGN.md §GroupBy kernels): LDS-dense for small ranges,
  // radix partition for the north-star range, global atomics as the wide
  // fallback (slow but correct for any range below the slot cap)
  auto route = [&](auto aopTag) -> int {
    constexpr int AOP = decltype(aopTag)::value;
    if (n > 0 && n_slots <= GB_DENSE_MAX)
      return gb_dense_path<AOP>(keys, ptrs, nvals, key_min, n_slots, sums,
                                rowcnt, counts, d_err);
    // sum-only uses 8192-key buckets (bigger scatter chunks; the u8-touch agg
    // table still fits 2 blocks/CU); count/mean use 4096 (3 blocks/CU with
    // the counts table)
    const bool want_cnt = counts != 0;
    const int64_t nb13 = (n_slots + (1 << 13) - 1) >> 13;
    const int64_t nb12 = (n_slots + (1 << 12) - 1) >> 12;
    const int rl_ok = (!want_cnt && nb13 <= GB_MAX_BUCKETS) ? 13
                      : (want_cnt && nb12 <= GB_MAX_BUCKETS) ? 12
                      : (nb13 <= GB_MAX_BUCKETS) ? 13 : 0;
    if (n > 0 && rl_ok) {
      // wide frames run the radix path in chunks of 2 value columns
      // (re-reading keys per chunk still beats the atomic fallback ~5x)
      int rc2 = HF_OK;
      for (int c0 = 0; c0 < (nvals ? nvals : 1) && rc2 == HF_OK; c0 += 2) {
        GbPtrs sub{};
        const int nv = nvals == 0 ? 0 : std::min(2, nvals - c0);
        for (int c = 0; c < nv; ++c) sub.vals[c] = ptrs.vals[c0 + c];
        const uintptr_t ssub = sums + (uintptr_t)c0 * n_slots * 8;
        const uintptr_t csub =
            counts ? counts + (uintptr_t)c0 * n_slots * 8 : 0;
        rc2 = rl_ok == 13
                  ? gb_radix_path<13, AOP>(const_cast<hf_col*>(keys), sub, nv,
                                           key_min, n_slots, ssub, rowcnt,
                                           csub, d_err, c0 == 0)
                  : gb_radix_path<12, AOP>(const_cast<hf_col*>(keys), sub, nv,
                                           key_min, n_slots, ssub, rowcnt,
                                           csub, d_err, c0 == 0);
        if (nvals == 0) break;
      }
      return rc2;
    }
    return -1;  // fall through to the atomic path
  };
  int routed = agg_op == HF_AGG_SUM ? route(std::integral_constant<int, HF_AGG_SUM>{})
               : agg_op == HF_AGG_MIN ? route(std::integral_constant<int, HF_AGG_MIN>{})
                                      : route(std::integral_constant<int, HF_AGG_MAX>{});
  if (routed != -1) return routed;
  auto launch = [&](auto nvTag, auto cntTag) {
    constexpr int NV = decltype(nvTag)::value;
    constexpr bool CNT = decltype(cntTag)::value;
    auto go = [&](auto aopTag) {
      constexpr int AOP = decltype(aopTag)::value;
      return timed_launch("gb_accum", [&] {
        hipLaunchKernelGGL((k_gb_accum<NV, CNT, AOP>),
                           dim3(grid_for((n >> 1) + 1)), dim3(BLOCK), 0,
                           g.stream, (const int64_t*)keys->dptr, ptrs, n,
                           key_min, n_slots, (double*)sums,
                           (unsigned long long*)rowcnt,
                           (unsigned long long*)counts, d_err);
      });
    };
    return agg_op == HF_AGG_SUM ? go(std::integral_constant<int, HF_AGG_SUM>{})
           : agg_op == HF_AGG_MIN ? go(std::integral_constant<int, HF_AGG_MIN>{})
                                  : go(std::integral_constant<int, HF_AGG_MAX>{});
  };
  const bool cnt = counts != 0;
  switch (nvals) {
#define HF_GB_CASE(NV)                                                        \
  case NV:                                                                    \
    return cnt ? launch(std::integral_constant<int, NV>{},                    \
                        std::integral_constant<bool, true>{})                 \
               : launch(std::integral_constant<int, NV>{},                    \
                        std::integral_constant<bool, false>{});
    HF_GB_CASE(1) HF_GB_CASE(2) HF_GB_CASE(3) HF_GB_CASE(4)
    HF_GB_CASE(5) HF_GB_CASE(6) HF_GB_CASE(7) HF_GB_CASE(8)
#undef HF_GB_CASE
    case 0:
      return cnt ? launch(std::integral_constant<int, 0>{},
                          std::integral_constant<bool, true>{})
                 : launch(std::integral_constant<int, 0>{},
                          std::integral_constant<bool, false>{});
  }
  return set_err(HF_ERR_ARG, "hf_groupby_accum", "nvals out of range");
}

int hf_groupby_compact(uintptr_t sums, uintptr_t rowcnt, uintptr_t counts,
                       int nvals, int64_t key_min, int64_t n_slots,
                       hf_col** out_keys, hf_col** out_sums, hf_col** out_counts,
                       int64_t* n_groups) {
  HF_NEED_INIT("hf_groupby_compact");
  if (!rowcnt || nvals < 0 || nvals > GB_MAX_VALS || !out_keys || !n_groups)
    return set_err(HF_ERR_ARG, "hf_groupby_compact", "bad args");
  const int64_t ntiles = (n_slots + COMPACT_TILE - 1) / COMPACT_TILE;
  // check the accumulate-phase error word
  unsigned long long h_err = 0;
  unsigned long long* d_err =
      (unsigned long long*)((char*)g.d_scratch + SCRATCH_GB_ERR);
  HF_HIP("hf_groupby_compact",
         hipMemcpyAsync(&h_err, d_err, 8, hipMemcpyDeviceToHost, g.stream));
  // tile counts
  int64_t* d_tiles = nullptr;
  HF_HIP("hf_groupby_compact",
         dev_alloc((void**)&d_tiles, (ntiles + 1) * 8, g.stream));
  int64_t* d_total = (int64_t*)((char*)g.d_scratch + SCRATCH_NGROUPS);
  int rc = timed_launch("gb_compact_count", [&] {
    hipLaunchKernelGGL(k_compact_count, dim3((uint32_t)ntiles), dim3(BLOCK), 0,
                       g.stream, (const unsigned long long*)rowcnt, n_slots, d_tiles);
  });
  if (rc != HF_OK) return rc;
  rc = timed_launch("gb_compact_scan", [&] {
    hipLaunchKernelGGL(k_compact_scan, dim3(1), dim3(1024), 0, g.stream, d_tiles,
                       ntiles, d_total);
  });
  if (rc != HF_OK) return rc;
  int64_t total = 0;
  HF_HIP("hf_groupby_compact",
         hipMemcpyAsync(&total, d_total, 8, hipMemcpyDeviceToHost, g.stream));
  HF_HIP("hf_groupby_compact", hipStreamSynchronize(g.stream));
  if (h_err != 0) {
    dev_free(d_tiles, g.stream);
    char buf[128];
    snprintf(buf, sizeof buf,
             "%llu keys fell outside [key_min, key_min+n_slots) during accumulate",
             h_err);
    hipMemsetAsync(d_err, 0, 8, g.stream);
    return set_err(HF_ERR_ARG, "hf_groupby_compact", buf);
  }
  // allocate outputs
  rc = hf_col_alloc(total, HF_INT64, out_keys);
  if (rc != HF_OK) return rc;
  std::vector<double*> h_sum_ptrs(nvals ? nvals : 1);
  std::vector<int64_t*> h_cnt_ptrs(nvals ? nvals : 1);
  for (int c = 0; c < nvals; ++c) {
    rc = hf_col_alloc(total, HF_FLOAT64, &out_sums[c]);
    if (rc != HF_OK) return rc;
    h_sum_ptrs[c] = (double*)out_sums[c]->dptr;
    if (counts && out_counts) {
      rc = hf_col_alloc(total, HF_INT64, &out_counts[c]);
      if (rc != HF_OK) return rc;
      h_cnt_ptrs[c] = (int64_t*)out_counts[c]->dptr;
    }
  }
  // ship the per-column output pointer arrays to the device
  double** d_sum_ptrs = nullptr;
  int64_t** d_cnt_ptrs = nullptr;
  HF_HIP("hf_groupby_compact",
         dev_alloc((void**)&d_sum_ptrs, sizeof(double*) * (nvals ? nvals : 1),
                        g.stream));
  HF_HIP("hf_groupby_compact",
         dev_alloc((void**)&d_cnt_ptrs, sizeof(int64_t*) * (nvals ? nvals : 1),
                        g.stream));
  HF_HIP("hf_groupby_compact",
         hipMemcpyAsync(d_sum_ptrs, h_sum_ptrs.data(), sizeof(double*) * nvals,
                        hipMemcpyHostToDevice, g.stream));
  if (counts && out_counts)
    HF_HIP("hf_groupby_compact",
           hipMemcpyAsync(d_cnt_ptrs, h_cnt_ptrs.data(), sizeof(int64_t*) * nvals,
                          hipMemcpyHostToDevice, g.stream));
  const bool want_counts = counts && out_counts;
  rc = timed_launch("gb_compact_scatter", [&] {
    if (want_counts)
      hipLaunchKernelGGL(k_compact_scatter<true>, dim3((uint32_t)ntiles), dim3(BLOCK),
                         0, g.stream, (const double*)sums,
                         (const unsigned long long*)rowcnt,
                         (const unsigned long long*)counts, nvals, key_min, n_slots,
                         d_tiles, (int64_t*)(*out_keys)->dptr, d_sum_ptrs, d_cnt_ptrs);
    else
      hipLaunchKernelGGL(k_compact_scatter<false>, dim3((uint32_t)ntiles), dim3(BLOCK),
                         0, g.stream, (const double*)sums,
                         (const unsigned long long*)rowcnt,
                         (const unsigned long long*)counts, nvals, key_min, n_slots,
                         d_tiles, (int64_t*)(*out_keys)->dptr, d_sum_ptrs, d_cnt_ptrs);
  });
  dev_free(d_tiles, g.stream);
  dev_free(d_sum_ptrs, g.stream);
  dev_free(d_cnt_ptrs, g.stream);
  if (rc != HF_OK) return rc;
  *n_groups = total;
  return HF_OK;
}

int hf_col_concat(const hf_col* const* cols, int ncols, hf_col** out) {
  HF_NEED_INIT("hf_col_concat");
  if (!cols || ncols <= 0 || !out)
    return set_err(HF_ERR_ARG, "hf_col_concat", "bad args");
  const int dt = cols[0]->dtype;
  int64_t total = 0;
  for (int i = 0; i < ncols; ++i) {
    if (!cols[i] || cols[i]->dtype != dt)
      return set_err(HF_ERR_ARG, "hf_col_concat", "dtype mismatch");
    total += cols[i]->len;
  }
  int rc = hf_col_alloc(total, dt, out);
  if (rc != HF_OK) return rc;
  char* dst = (char*)(*out)->dptr;
  const int64_t esz = dtype_size(dt);
  for (int i = 0; i < ncols; ++i) {
    const int64_t bytes = cols[i]->len * esz;
    if (bytes > 0)
      HF_HIP("hf_col_concat",
             hipMemcpyAsync(dst, cols[i]->dptr, bytes,
                            hipMemcpyDeviceToDevice, g.stream));
    dst += bytes;
  }
  return HF_OK;
}

int hf_fill_i64(uintptr_t dptr, int64_t value, int64_t n) {
  HF_NEED_INIT("hf_fill_i64");
  if (n <= 0) return HF_OK;
  return timed_launch("fill_i64", [&] {
    hipLaunchKernelGGL(k_fill_i64, dim3((uint32_t)grid_for(n)), dim3(BLOCK), 0,
                       g.stream, (int64_t*)dptr, value, n);
  });
}

int hf_fill_randint(hf_col* col, uint64_t seed, int64_t lo, int64_t hi) {
  HF_NEED_INIT("hf_fill_randint");
  if (!col || col->dtype != HF_INT64 || hi <= lo)
    return set_err(HF_ERR_ARG, "hf_fill_randint", "int64 col, hi > lo");
  if (col->len == 0) return HF_OK;
  return timed_launch("fill_randint", [&] {
    hipLaunchKernelGGL(k_fill_randint, dim3((uint32_t)grid_for(col->len)),
                       dim3(BLOCK), 0, g.stream, (int64_t*)col->dptr,
                       col->len, seed, lo, (uint64_t)(hi - lo));
  });
}

int hf_fill_randf64(hf_col* col, uint64_t seed) {
  HF_NEED_INIT("hf_fill_randf64");
  if (!col || col->dtype != HF_FLOAT64)
    return set_err(HF_ERR_ARG, "hf_fill_randf64", "float64 col");
  if (col->len == 0) return HF_OK;
  return timed_launch("fill_randf64", [&] {
    hipLaunchKernelGGL(k_fill_randf64, dim3((uint32_t)grid_for(col->len)),
                       dim3(BLOCK), 0, g.stream, (double*)col->dptr,
                       col->len, seed);
  });
}

int hf_fill_randcdf(hf_col* col, uint64_t seed, const hf_col* cdf) {
  HF_NEED_INIT("hf_fill_randcdf");
  if (!col || col->dtype != HF_INT64 || !cdf || cdf->dtype != HF_FLOAT64 ||
      cdf->len <= 0)
    return set_err(HF_ERR_ARG, "hf_fill_randcdf",
                   "int64 col, non-empty float64 cdf");
  if (col->len == 0) return HF_OK;
  return timed_launch("fill_randcdf", [&] {
    hipLaunchKernelGGL(k_fill_randcdf, dim3((uint32_t)grid_for(col->len)),
                       dim3(BLOCK), 0, g.stream, (int64_t*)col->dptr,
                       col->len, seed, (const double*)cdf->dptr, cdf->len);
  });
}

int hf_groupby_hash_accum(const hf_col* keys, const hf_col* const* vals,
                          int nvals, int agg_op, int64_t H,
                          uintptr_t tkey, uintptr_t sums, uintptr_t rowcnt,
                          uintptr_t counts) {
  HF_NEED_INIT("hf_groupby_hash_accum");
  if (!keys || nvals < 0 || nvals > GB_MAX_VALS || H < 2 || (H & (H - 1)))
    return set_err(HF_ERR_ARG, "hf_groupby_hash_accum",
                   "bad args (H power of 2, nvals<=8)");
  if (keys->dtype != HF_INT64)
    return set_err(HF_ERR_ARG, "hf_groupby_hash_accum", "keys must be int64");
  GbPtrs ptrs{};
  for (int c = 0; c < nvals; ++c) {
    if (!vals[c] || vals[c]->dtype != HF_FLOAT64 || vals[c]->len != keys->len)
      return set_err(HF_ERR_ARG, "hf_groupby_hash_accum",
                     "vals must be float64 columns of keys' length");
    ptrs.vals[c] = (const double*)vals[c]->dptr;
  }
  const int64_t n = keys->len;
  if (n == 0) return HF_OK;
  unsigned long long* d_full =
      (unsigned long long*)((char*)g.d_scratch + SCRATCH_HASH_FULL);
  const bool cnt = counts != 0;
  auto launch = [&](auto nvTag, auto cTag, auto aTag) {
    constexpr int NV = decltype(nvTag)::value;
    constexpr bool C = decltype(cTag)::value;
    constexpr int A = decltype(aTag)::value;
    return timed_launch("gb_hash_accum", [&] {
      hipLaunchKernelGGL((k_gb_hash_accum<NV, C, A>),
                         dim3((uint32_t)grid_for(n)), dim3(BLOCK), 0, g.stream,
                         (const int64_t*)keys->dptr, ptrs, n, H,
                         (long long*)tkey, (double*)sums,
                         (unsigned long long*)rowcnt,
                         (unsigned long long*)counts, d_full);
    });
  };
  auto withA = [&](auto nvTag, auto cTag) {
    return agg_op == HF_AGG_SUM
               ? launch(nvTag, cTag, std::integral_constant<int, HF_AGG_SUM>{})
           : agg_op == HF_AGG_MIN
               ? launch(nvTag, cTag, std::integral_constant<int, HF_AGG_MIN>{})
               : launch(nvTag, cTag,
                        std::integral_constant<int, HF_AGG_MAX>{});
  };
  switch (nvals) {
#define HF_GH_CASE(NV)                                                          case NV:                                                                        return cnt ? withA(std::integral_constant<int, NV>{},                                            std::integral_constant<bool, true>{})                                 : withA(std::integral_constant<int, NV>{},                                            std::integral_constant<bool, false>{});
    HF_GH_CASE(0) HF_GH_CASE(1) HF_GH_CASE(2) HF_GH_CASE(3) HF_GH_CASE(4)
    HF_GH_CASE(5) HF_GH_CASE(6) HF_GH_CASE(7) HF_GH_CASE(8)
#undef HF_GH_CASE
  }
  return set_err(HF_ERR_ARG, "hf_groupby_hash_accum", "nvals out of range");
}

int hf_groupby_hash_compact(uintptr_t tkey, uintptr_t sums, uintptr_t rowcnt,
                            uintptr_t counts, int nvals, int64_t H,
                            hf_col** out_keys, hf_col** out_sums,
                            hf_col** out_counts, int64_t* n_groups) {
  HF_NEED_INIT("hf_groupby_hash_compact");
  if (!tkey || !rowcnt || nvals < 0 || nvals > GB_MAX_VALS || !out_keys ||
      !n_groups)
    return set_err(HF_ERR_ARG, "hf_groupby_hash_compact", "bad args");
  // surface "table full" from the accumulate phase
  unsigned long long h_full = 0;
  unsigned long long* d_full =
      (unsigned long long*)((char*)g.d_scratch + SCRATCH_HASH_FULL);
  HF_HIP("hf_groupby_hash_compact",
         hipMemcpyAsync(&h_full, d_full, 8, hipMemcpyDeviceToHost, g.stream));
  HF_HIP("hf_groupby_hash_compact", hipStreamSynchronize(g.stream));
  if (h_full) {
    hipMemsetAsync(d_full, 0, 8, g.stream);
    return set_err(HF_ERR_UNSUPPORTED, "hf_groupby_hash_compact",
                   "hash table full — retry with a larger H");
  }
  const int64_t L = H + 1;
  const int gpu = g.gpu;
  // filter present slots (rowcnt doubles as the nonzero mask)
  hf_col mask_view{(void*)rowcnt, L, HF_INT64, gpu};
  hf_filterplan* plan = nullptr;
  int64_t kept = 0;
  int rc = hf_filter_plan(&mask_view, &plan, &kept);
  if (rc != HF_OK) return rc;
  hf_col tkey_view{(void*)tkey, L, HF_INT64, gpu};
  hf_col* ckeys = nullptr;
  rc = hf_filter_apply(plan, &tkey_view, &ckeys);
  std::vector<hf_col*> csums((size_t)(nvals ? nvals : 1), nullptr);
  std::vector<hf_col*> ccnts((size_t)(nvals ? nvals : 1), nullptr);
  for (int c = 0; c < nvals && rc == HF_OK; ++c) {
    hf_col sv{(char*)sums + (int64_t)c * L * 8, L, HF_FLOAT64, gpu};
    rc = hf_filter_apply(plan, &sv, &csums[c]);
    if (rc == HF_OK && counts) {
      hf_col cv{(char*)counts + (int64_t)c * L * 8, L, HF_INT64, gpu};
      rc = hf_filter_apply(plan, &cv, &ccnts[c]);
    }
  }
  hf_filter_plan_free(plan);
  // sort surviving keys (wide radix handles any int64 range) and gather
  hf_col* perm = nullptr;
  if (rc == HF_OK) rc = hf_sort_perm(ckeys, 1, &perm);
  if (rc == HF_OK) rc = hf_gather(ckeys, perm, out_keys);
  for (int c = 0; c < nvals && rc == HF_OK; ++c) {
    rc = hf_gather(csums[c], perm, &out_sums[c]);
    if (rc == HF_OK && counts && out_counts)
      rc = hf_gather(ccnts[c], perm, &out_counts[c]);
  }
  hf_col_free(ckeys);
  hf_col_free(perm);
  for (int c = 0; c < nvals; ++c) {
    hf_col_free(csums[c]);
    hf_col_free(ccnts[c]);
  }
  if (rc != HF_OK) return rc;
  *n_groups = kept;
  return HF_OK;
}

static const int64_t* plan_tiles(const hf_filterplan* p);

int hf_cumsum(const hf_col* col, int agg_op, hf_col** out) {
  HF_NEED_INIT("hf_cumsum");
  if (!col || !out) return set_err(HF_ERR_ARG, "hf_cumsum", "null");
  if (agg_op != HF_AGG_SUM && agg_op != HF_AGG_MIN &&
      agg_op != HF_AGG_MAX && agg_op != HF_AGG_PROD)
    return set_err(HF_ERR_ARG, "hf_cumsum", "bad agg_op");
  const int64_t n = col->len;
  int rc = hf_col_alloc(n, col->dtype, out);
  if (rc != HF_OK) return rc;
  if (n == 0) return HF_OK;
  const int64_t ntiles = (n + FILT_TILE - 1) / FILT_TILE;
  void* d_ts = nullptr;
  HF_HIP("hf_cumsum", dev_alloc(&d_ts, ntiles * 8, g.stream));
  auto run = [&](auto tTag, auto opTag) -> int {
    using T = decltype(tTag);
    constexpr int OP = decltype(opTag)::value;
    int r2 = timed_launch("cumsum_tiles", [&] {
      hipLaunchKernelGGL((k_cumsum_tiles<T, OP>), dim3((uint32_t)ntiles),
                         dim3(BLOCK), 0, g.stream, (const T*)col->dptr, n,
                         (T*)d_ts);
    });
    if (r2 != HF_OK) return r2;
    r2 = timed_launch("cumsum_scan", [&] {
      hipLaunchKernelGGL((k_cumsum_scan_tiles<T, OP>), dim3(1), dim3(1024),
                         0, g.stream, (T*)d_ts, ntiles);
    });
    if (r2 != HF_OK) return r2;
    return timed_launch("cumsum_apply", [&] {
      hipLaunchKernelGGL((k_cumsum_apply<T, OP>), dim3((uint32_t)ntiles),
                         dim3(BLOCK), 0, g.stream, (const T*)col->dptr, n,
                         (const T*)d_ts, (T*)(*out)->dptr);
    });
  };
  auto runT = [&](auto opTag) -> int {
    return (col->dtype == HF_FLOAT64) ? run(double{}, opTag)
                                      : run(int64_t{}, opTag);
  };
  rc = agg_op == HF_AGG_MIN
           ? runT(std::integral_constant<int, HF_AGG_MIN>{})
       : agg_op == HF_AGG_MAX
           ? runT(std::integral_constant<int, HF_AGG_MAX>{})
       : agg_op == HF_AGG_PROD
           ? runT(std::integral_constant<int, HF_AGG_PROD>{})
           : runT(std::integral_constant<int, HF_AGG_SUM>{});
  dev_free(d_ts, g.stream);
  if (rc != HF_OK) { hf_col_free(*out); *out = nullptr; }
  return rc;
}

int hf_seg_cumsum(const hf_col* col, const hf_col* heads, int agg_op,
                  hf_col** out) {
  HF_NEED_INIT("hf_seg_cumsum");
  if (!col || !heads || !out)
    return set_err(HF_ERR_ARG, "hf_seg_cumsum", "null");
  if (heads->dtype != HF_INT64 || heads->len != col->len)
    return set_err(HF_ERR_ARG, "hf_seg_cumsum",
                   "heads must be an int64 0/1 column of the same length");
  if (agg_op != HF_AGG_SUM && agg_op != HF_AGG_MIN &&
      agg_op != HF_AGG_MAX && agg_op != HF_AGG_PROD)
    return set_err(HF_ERR_ARG, "hf_seg_cumsum", "bad agg_op");
  const int64_t n = col->len;
  int rc = hf_col_alloc(n, col->dtype, out);
  if (rc != HF_OK) return rc;
  if (n == 0) return HF_OK;
  const int64_t ntiles = (n + FILT_TILE - 1) / FILT_TILE;
  void* d_tv = nullptr;
  void* d_tf = nullptr;
  rc = [&]() -> int {
    // both tile allocs inside the lambda so the hf_col_free(*out) error
    // path below covers an alloc failure (no *out leak)
    HF_HIP("hf_seg_cumsum", dev_alloc(&d_tv, ntiles * 8, g.stream));
    HF_HIP("hf_seg_cumsum", dev_alloc(&d_tf, ntiles * 4, g.stream));
    auto run = [&](auto tTag, auto opTag) -> int {
      using T = decltype(tTag);
      constexpr int OP = decltype(opTag)::value;
      int r2 = timed_launch("seg_tiles", [&] {
        hipLaunchKernelGGL((k_seg_tiles<T, OP>), dim3((uint32_t)ntiles),
                           dim3(BLOCK), 0, g.stream, (const T*)col->dptr,
                           (const int64_t*)heads->dptr, n, (T*)d_tv,
                           (unsigned*)d_tf);
      });
      if (r2 != HF_OK) return r2;
      r2 = timed_launch("seg_scan", [&] {
        hipLaunchKernelGGL((k_seg_scan_tiles<T, OP>), dim3(1), dim3(1024),
                           0, g.stream, (T*)d_tv, (unsigned*)d_tf, ntiles);
      });
      if (r2 != HF_OK) return r2;
      return timed_launch("seg_apply", [&] {
        hipLaunchKernelGGL((k_seg_apply<T, OP>), dim3((uint32_t)ntiles),
                           dim3(BLOCK), 0, g.stream, (const T*)col->dptr,
                           (const int64_t*)heads->dptr, n, (const T*)d_tv,
                           (const unsigned*)d_tf, (T*)(*out)->dptr);
      });
    };
    auto runT = [&](auto opTag) -> int {
      return (col->dtype == HF_FLOAT64) ? run(double{}, opTag)
                                        : run(int64_t{}, opTag);
    };
    return agg_op == HF_AGG_MIN
               ? runT(std::integral_constant<int, HF_AGG_MIN>{})
           : agg_op == HF_AGG_MAX
               ? runT(std::integral_constant<int, HF_AGG_MAX>{})
           : agg_op == HF_AGG_PROD
               ? runT(std::integral_constant<int, HF_AGG_PROD>{})
               : runT(std::integral_constant<int, HF_AGG_SUM>{});
  }();
  if (d_tf) dev_free(d_tf, g.stream);
  if (d_tv) dev_free(d_tv, g.stream);
  if (rc != HF_OK) { hf_col_free(*out); *out = nullptr; }
  return rc;
}

int hf_cross_idx(int64_t nl, int64_t nr, hf_col** lidx, hf_col** ridx) {
  HF_NEED_INIT("hf_cross_idx");
  if (!lidx || !ridx || nl < 0 || nr <= 0)
    return set_err(HF_ERR_ARG, "hf_cross_idx", "bad args");
  const int64_t n = nl * nr;
  int rc = hf_col_alloc(n, HF_INT64, lidx);
  if (rc != HF_OK) return rc;
  rc = hf_col_alloc(n, HF_INT64, ridx);
  if (rc != HF_OK) { hf_col_free(*lidx); *lidx = nullptr; return rc; }
  if (n == 0) return HF_OK;
  rc = timed_launch("cross_idx", [&] {
    hipLaunchKernelGGL(k_cross_idx, dim3((uint32_t)grid_for(n)), dim3(BLOCK),
                       0, g.stream, n, nr, (int64_t*)(*lidx)->dptr,
                       (int64_t*)(*ridx)->dptr);
  });
  if (rc != HF_OK) {
    hf_col_free(*lidx); hf_col_free(*ridx);
    *lidx = *ridx = nullptr;
  }
  return rc;
}

int hf_scatter(const hf_col* col, const hf_col* idx, hf_col** out) {
  HF_NEED_INIT("hf_scatter");
  if (!col || !idx || !out) return set_err(HF_ERR_ARG, "hf_scatter", "null");
  if (idx->dtype != HF_INT64 || idx->len != col->len)
    return set_err(HF_ERR_ARG, "hf_scatter",
                   "index must be an int64 permutation of the column");
  int rc = hf_col_alloc(col->len, col->dtype, out);
  if (rc != HF_OK) return rc;
  const int64_t n = col->len;
  if (n == 0) return HF_OK;
  rc = timed_launch("scatter", [&] {
    if (col->dtype == HF_FLOAT64)
      hipLaunchKernelGGL((k_scatter<double>), dim3((uint32_t)grid_for(n)),
                         dim3(BLOCK), 0, g.stream, (const double*)col->dptr,
                         (const int64_t*)idx->dptr, (double*)(*out)->dptr, n);
    else
      hipLaunchKernelGGL((k_scatter<int64_t>), dim3((uint32_t)grid_for(n)),
                         dim3(BLOCK), 0, g.stream, (const int64_t*)col->dptr,
                         (const int64_t*)idx->dptr, (int64_t*)(*out)->dptr,
                         n);
  });
  if (rc != HF_OK) { hf_col_free(*out); *out = nullptr; }
  return rc;
}

int hf_ordered_i64(const hf_col* col, int direction, hf_col** out) {
  HF_NEED_INIT("hf_ordered_i64");
  if (!col || !out) return set_err(HF_ERR_ARG, "hf_ordered_i64", "null");
  const int want = direction ? HF_INT64 : HF_FLOAT64;
  if (col->dtype != want)
    return set_err(HF_ERR_ARG, "hf_ordered_i64",
                   direction ? "inverse needs an int64 column"
                             : "forward needs a float64 column");
  const int64_t n = col->len;
  int rc = hf_col_alloc(n, direction ? HF_FLOAT64 : HF_INT64, out);
  if (rc != HF_OK) return rc;
  if (n > 0) {
    rc = timed_launch("f64_ordered", [&] {
      if (direction)
        hipLaunchKernelGGL(k_ordered_f64, dim3((uint32_t)grid_for(n)),
                           dim3(BLOCK), 0, g.stream,
                           (const int64_t*)col->dptr, (double*)(*out)->dptr,
                           n);
      else
        hipLaunchKernelGGL(k_f64_ordered, dim3((uint32_t)grid_for(n)),
                           dim3(BLOCK), 0, g.stream,
                           (const double*)col->dptr,
                           (int64_t*)(*out)->dptr, n);
    });
    if (rc != HF_OK) { hf_col_free(*out); *out = nullptr; }
  }
  return rc;
}

int hf_search_sorted(const hf_col* keys, const hf_col* sorted_uniq,
                     hf_col** out) {
  HF_NEED_INIT("hf_search_sorted");
  if (!keys || !sorted_uniq || !out)
    return set_err(HF_ERR_ARG, "hf_search_sorted", "null");
  if (keys->dtype != HF_INT64 || sorted_uniq->dtype != HF_INT64)
    return set_err(HF_ERR_ARG, "hf_search_sorted", "int64 columns required");
  const int64_t n = keys->len;
  int rc = hf_col_alloc(n, HF_INT64, out);
  if (rc != HF_OK) return rc;
  if (n > 0) {
    rc = timed_launch("search_sorted", [&] {
      hipLaunchKernelGGL(k_search_sorted, dim3((uint32_t)grid_for(n)),
                         dim3(BLOCK), 0, g.stream,
                         (const int64_t*)keys->dptr, n,
                         (const int64_t*)sorted_uniq->dptr,
                         sorted_uniq->len, (int64_t*)(*out)->dptr);
    });
    if (rc != HF_OK) { hf_col_free(*out); *out = nullptr; }
  }
  return rc;
}

int hf_shuffle_dest(const hf_col* keys, const int64_t* splitters, int nsplit,
                    hf_col** dest) {
  HF_NEED_INIT("hf_shuffle_dest");
  if (!keys || !dest || nsplit < 0 || nsplit > 63 ||
      (nsplit && !splitters))
    return set_err(HF_ERR_ARG, "hf_shuffle_dest", "bad args");
  if (keys->dtype != HF_INT64)
    return set_err(HF_ERR_ARG, "hf_shuffle_dest", "keys must be int64");
  const int64_t n = keys->len;
  int rc = hf_col_alloc(n, HF_INT64, dest);
  if (rc != HF_OK) return rc;
  int64_t* d_split = nullptr;
  if (nsplit) {
    HF_HIP("hf_shuffle_dest", dev_alloc((void**)&d_split, nsplit * 8, g.stream));
    HF_HIP("hf_shuffle_dest",
           hipMemcpyAsync(d_split, splitters, nsplit * 8,
                          hipMemcpyHostToDevice, g.stream));
  }
  if (n > 0) {
    rc = timed_launch("shuffle_dest", [&] {
      hipLaunchKernelGGL(k_shuffle_dest, dim3((uint32_t)grid_for(n)),
                         dim3(BLOCK), 0, g.stream,
                         (const int64_t*)keys->dptr, d_split, nsplit,
                         (long long*)(*dest)->dptr, n);
    });
  }
  if (d_split) {
    // splitters are host memory borrowed only until the H2D lands
    hipStreamSynchronize(g.stream);
    dev_free(d_split, g.stream);
  }
  if (rc != HF_OK) { hf_col_free(*dest); *dest = nullptr; }
  return rc;
}

int hf_memcpy_dd(uintptr_t dst, uintptr_t src, int64_t bytes) {
  HF_NEED_INIT("hf_memcpy_dd");
  if (bytes < 0 || (bytes && (!dst || !src)))
    return set_err(HF_ERR_ARG, "hf_memcpy_dd", "bad args");
  if (bytes)
    HF_HIP("hf_memcpy_dd",
           hipMemcpyAsync((void*)dst, (void*)src, (size_t)bytes,
                          hipMemcpyDeviceToDevice, g.stream));
  return HF_OK;
}

int hf_groupby_sorted(const hf_col* sorted_keys, const hf_col* const* vals,
                      int nvals, int agg_op, int want_counts,
                      hf_col** out_keys, hf_col** out_sums,
                      hf_col** out_counts, int64_t* n_groups) {
  HF_NEED_INIT("hf_groupby_sorted");
  if (!sorted_keys || nvals < 0 || nvals > GB_MAX_VALS || !out_keys ||
      !n_groups)
    return set_err(HF_ERR_ARG, "hf_groupby_sorted", "bad args");
  if (sorted_keys->dtype != HF_INT64)
    return set_err(HF_ERR_ARG, "hf_groupby_sorted", "keys must be int64");
  for (int c = 0; c < nvals; ++c)
    if (!vals[c] || vals[c]->dtype != HF_FLOAT64 ||
        vals[c]->len != sorted_keys->len)
      return set_err(HF_ERR_ARG, "hf_groupby_sorted",
                     "vals must be float64 columns of keys' length");
  const int64_t n = sorted_keys->len;
  // head flags -> filter plan (run count + tile bases)
  hf_col* head = nullptr;
  int rc = hf_col_alloc(n, HF_INT64, &head);
  if (rc != HF_OK) return rc;
  if (n > 0) {
    rc = timed_launch("gb_sorted_heads", [&] {
      hipLaunchKernelGGL(k_head_flags, dim3((uint32_t)grid_for(n)),
                         dim3(BLOCK), 0, g.stream,
                         (const int64_t*)sorted_keys->dptr, n,
                         (long long*)head->dptr);
    });
    if (rc != HF_OK) { hf_col_free(head); return rc; }
  }
  hf_filterplan* plan = nullptr;
  int64_t C = 0;
  rc = hf_filter_plan(head, &plan, &C);
  if (rc != HF_OK) { hf_col_free(head); return rc; }
  // outputs: unique keys (filter of sorted keys) + per-run aggregates
  rc = hf_filter_apply(plan, sorted_keys, out_keys);
  const bool cnt = want_counts != 0;
  // per-run aggregate buffers
  const int64_t Ca = C > 0 ? C : 1;
  double* gsums = nullptr;
  unsigned long long* growcnt = nullptr;
  unsigned long long* gcounts = nullptr;
  if (rc == HF_OK)
    HF_HIP("hf_groupby_sorted",
           dev_alloc((void**)&gsums, Ca * 8 * (nvals ? nvals : 1), g.stream));
  if (rc == HF_OK)
    HF_HIP("hf_groupby_sorted", dev_alloc((void**)&growcnt, Ca * 8, g.stream));
  if (rc == HF_OK && cnt)
    HF_HIP("hf_groupby_sorted",
           dev_alloc((void**)&gcounts, Ca * 8 * (nvals ? nvals : 1), g.stream));
  if (rc == HF_OK) {
    const double init =
        agg_op == HF_AGG_SUM ? 0.0
        : agg_op == HF_AGG_MIN ? __builtin_huge_val() : -__builtin_huge_val();
    if (nvals) {
      rc = timed_launch("fill_f64", [&] {
        hipLaunchKernelGGL(k_fill_f64,
                           dim3((uint32_t)grid_for(Ca * nvals)), dim3(BLOCK),
                           0, g.stream, gsums, init, Ca * nvals);
      });
    }
    HF_HIP("hf_groupby_sorted", hipMemsetAsync(growcnt, 0, Ca * 8, g.stream));
    if (cnt && nvals)
      HF_HIP("hf_groupby_sorted",
             hipMemsetAsync(gcounts, 0, Ca * 8 * nvals, g.stream));
  }
  // aggregate per column (ROWCNT on the first launch only)
  const int64_t ntiles = n > 0 ? (n + FILT_TILE - 1) / FILT_TILE : 1;
  auto seg = [&](auto aTag, auto cTag, auto rTag, auto vTag, const double* v,
                 double* gs, unsigned long long* gc) {
    constexpr int A = decltype(aTag)::value;
    constexpr bool CC = decltype(cTag)::value;
    constexpr bool R = decltype(rTag)::value;
    constexpr bool V = decltype(vTag)::value;
    return timed_launch("gb_segagg", [&] {
      hipLaunchKernelGGL((k_segagg<A, CC, R, V>), dim3((uint32_t)ntiles),
                         dim3(BLOCK), 0, g.stream, (const long long*)head->dptr,
                         v, n, plan_tiles(plan), gs, growcnt, gc);
    });
  };
  using T = std::true_type;
  using F = std::false_type;
  auto segA = [&](auto cTag, auto rTag, auto vTag, const double* v, double* gs,
                  unsigned long long* gc) {
    return agg_op == HF_AGG_SUM
               ? seg(std::integral_constant<int, HF_AGG_SUM>{}, cTag, rTag,
                     vTag, v, gs, gc)
           : agg_op == HF_AGG_MIN
               ? seg(std::integral_constant<int, HF_AGG_MIN>{}, cTag, rTag,
                     vTag, v, gs, gc)
               : seg(std::integral_constant<int, HF_AGG_MAX>{}, cTag, rTag,
                     vTag, v, gs, gc);
  };
  if (rc == HF_OK && n > 0) {
    if (nvals == 0) {
      rc = segA(F{}, T{}, F{}, nullptr, nullptr, nullptr);
    } else {
      for (int c = 0; c < nvals && rc == HF_OK; ++c) {
        double* gs = gsums + (int64_t)c * Ca;
        unsigned long long* gc = cnt ? gcounts + (int64_t)c * Ca : nullptr;
        const double* v = (const double*)vals[c]->dptr;
        if (c == 0)
          rc = cnt ? segA(T{}, T{}, T{}, v, gs, gc)
                   : segA(F{}, T{}, T{}, v, gs, gc);
        else
          rc = cnt ? segA(T{}, F{}, T{}, v, gs, gc)
                   : segA(F{}, F{}, T{}, v, gs, gc);
      }
    }
  }
  // wrap aggregate buffers as columns (transfer ownership)
  for (int c = 0; c < nvals && rc == HF_OK; ++c) {
    hf_col* sc = new hf_col{};
    sc->len = C;
    sc->dtype = HF_FLOAT64;
    sc->gpu = g.gpu;
    HF_HIP("hf_groupby_sorted", dev_alloc(&sc->dptr, Ca * 8, g.stream));
    HF_HIP("hf_groupby_sorted",
           hipMemcpyAsync(sc->dptr, gsums + (int64_t)c * Ca, Ca * 8,
                          hipMemcpyDeviceToDevice, g.stream));
    out_sums[c] = sc;
    if (cnt && out_counts) {
      hf_col* cc2 = new hf_col{};
      cc2->len = C;
      cc2->dtype = HF_INT64;
      cc2->gpu = g.gpu;
      HF_HIP("hf_groupby_sorted", dev_alloc(&cc2->dptr, Ca * 8, g.stream));
      HF_HIP("hf_groupby_sorted",
             hipMemcpyAsync(cc2->dptr, gcounts + (int64_t)c * Ca, Ca * 8,
                            hipMemcpyDeviceToDevice, g.stream));
      out_counts[c] = cc2;
    }
  }
  hf_filter_plan_free(plan);
  hf_col_free(head);
  if (gsums) dev_free(gsums, g.stream);
  if (growcnt) dev_free(growcnt, g.stream);
  if (gcounts) dev_free(gcounts, g.stream);
  if (rc != HF_OK) return rc;
  *n_groups = C;
  return HF_OK;
}

int hf_col_slice(const hf_col* col, int64_t start, int64_t len, hf_col** out) {
  HF_NEED_INIT("hf_col_slice");
  if (!col || !out || start < 0 || len < 0 || start + len > col->len)
    return set_err(HF_ERR_ARG, "hf_col_slice", "bad range");
  int rc = hf_col_alloc(len, col->dtype, out);
  if (rc != HF_OK) return rc;
  const int64_t esz = dtype_size(col->dtype);
  if (len > 0)
    HF_HIP("hf_col_slice",
           hipMemcpyAsync((*out)->dptr, (const char*)col->dptr + start * esz,
                          len * esz, hipMemcpyDeviceToDevice, g.stream));
  return HF_OK;
}

// ---- join ----

int hf_join_build(const hf_col* rkeys, const hf_col* const* rvals, int nr,
                  int64_t key_min, int64_t n_slots, hf_join** out) {
  HF_NEED_INIT("hf_join_build");
  if (!rkeys || nr < 0 || nr > GB_MAX_VALS || !out || n_slots <= 0)
    return set_err(HF_ERR_ARG, "hf_join_build", "bad args (nr<=8)");
  if (rkeys->dtype != HF_INT64)
    return set_err(HF_ERR_ARG, "hf_join_build", "right keys must be int64");
  if (n_slots > (1LL << 27))
    return set_err(HF_ERR_UNSUPPORTED, "hf_join_build",
                   "right key range too large for the dense-range CSR "
                   "(hash build is a later round)");
  JoinConstPtrs rv{};
  for (int c = 0; c < nr; ++c) {
    // the join only MOVES right payloads (8 B opaque), so both dtypes pass
    if (!rvals[c] || dtype_size(rvals[c]->dtype) != 8 ||
        rvals[c]->len != rkeys->len)
      return set_err(HF_ERR_ARG, "hf_join_build",
                     "right vals must be 8-byte columns of keys' length");
    rv.vals[c] = (const double*)rvals[c]->dptr;
  }
  const int64_t n = rkeys->len;
  unsigned long long* d_hist_err =
      (unsigned long long*)((char*)g.d_scratch + SCRATCH_JOIN_HIST_ERR);
  unsigned long long* d_fix_err =
      (unsigned long long*)((char*)g.d_scratch + SCRATCH_JOIN_FIXUP_ERR);
  int64_t* d_total = (int64_t*)((char*)g.d_scratch + SCRATCH_NGROUPS);
  unsigned* d_cnt = nullptr;
  int64_t* d_tiles = nullptr;
  unsigned long long* d_csr = nullptr;
  const int64_t ntiles = (n_slots + JOIN_TILE - 1) / JOIN_TILE;
  HF_HIP("hf_join_build", dev_alloc((void**)&d_cnt, n_slots * 4, g.stream));
  HF_HIP("hf_join_build", hipMemsetAsync(d_cnt, 0, n_slots * 4, g.stream));
  HF_HIP("hf_join_build",
         dev_alloc((void**)&d_tiles, ntiles * 8, g.stream));
  HF_HIP("hf_join_build",
         dev_alloc((void**)&d_csr, (n_slots + 1) * 8, g.stream));
  int rc = timed_launch("join_hist", [&] {
    hipLaunchKernelGGL(k_hist_u32, dim3((uint32_t)grid_for(n)), dim3(BLOCK), 0,
                       g.stream, (const int64_t*)rkeys->dptr, n, key_min,
                       n_slots, d_cnt, d_hist_err);
  });
  if (rc != HF_OK) return rc;
  rc = timed_launch("join_scan", [&] {
    hipLaunchKernelGGL(k_tile_sums_u32, dim3((uint32_t)ntiles), dim3(BLOCK), 0,
                       g.stream, d_cnt, n_slots, d_tiles);
    hipLaunchKernelGGL(k_compact_scan, dim3(1), dim3(1024), 0, g.stream,
                       d_tiles, ntiles, d_total);
    hipLaunchKernelGGL(k_scan_apply_u32, dim3((uint32_t)ntiles), dim3(BLOCK),
                       0, g.stream, d_cnt, n_slots, d_tiles, d_total, d_csr);
  });
  if (rc != HF_OK) return rc;
  if (getenv("HF_JOIN_DEBUG")) {
    // host-side CSR verification (debug only): monotone, steps match the
    // histogram, total == rows
    std::vector<unsigned> h_cnt((size_t)n_slots);
    std::vector<unsigned long long> h_csr((size_t)n_slots + 1);
    hipMemcpyAsync(h_cnt.data(), d_cnt, n_slots * 4, hipMemcpyDeviceToHost,
                   g.stream);
    hipMemcpyAsync(h_csr.data(), d_csr, (n_slots + 1) * 8,
                   hipMemcpyDeviceToHost, g.stream);
    hipStreamSynchronize(g.stream);
    unsigned long long run = 0;
    int bad = 0;
    for (int64_t s = 0; s < n_slots && bad < 5; ++s) {
      if (h_csr[s] != run) {
        fprintf(stderr,
                "[join_debug] csr[%lld]=%llu expect %llu (cnt[s]=%u, "
                "tile=%lld)\n",
                (long long)s, h_csr[s], run, h_cnt[s], (long long)(s / 4096));
        ++bad;
      }
      run += h_cnt[s];
    }
    if (h_csr[n_slots] != run)
      fprintf(stderr, "[join_debug] csr[n]=%llu expect %llu\n", h_csr[n_slots],
              run);
    unsigned long long mx = 0;
    for (int64_t s = 0; s < n_slots; ++s)
      if (h_cnt[s] > mx) mx = h_cnt[s];
    fprintf(stderr, "[join_debug] hist total=%llu rows=%lld max_mult=%llu "
            "bad=%d\n", run, (long long)n, mx, bad);
  }
  // reuse d_cnt as the fill cursor
  HF_HIP("hf_join_build", hipMemsetAsync(d_cnt, 0, n_slots * 4, g.stream));
  hf_join* j = new hf_join{};
  j->key_min = key_min;
  j->n_slots = n_slots;
  j->n_right = n;
  j->nr = nr;
  j->d_csr = d_csr;
  for (int c = 0; c < nr; ++c) j->dtypes[c] = rvals[c]->dtype;
  const int64_t alloc_n = n > 0 ? n : 1;
  HF_HIP("hf_join_build",
         dev_alloc((void**)&j->d_jidx, alloc_n * 4, g.stream));
  JoinPtrs jv{};
  for (int c = 0; c < nr; ++c) {
    HF_HIP("hf_join_build",
           dev_alloc((void**)&j->d_jval[c], alloc_n * 8, g.stream));
    jv.vals[c] = j->d_jval[c];
  }
  auto fill = [&](auto nrTag) {
    constexpr int NR = decltype(nrTag)::value;
    int r2 = timed_launch("join_fill", [&] {
      hipLaunchKernelGGL((k_join_fill<NR>), dim3((uint32_t)grid_for(n)),
                         dim3(BLOCK), 0, g.stream, (const int64_t*)rkeys->dptr,
                         rv, n, key_min, n_slots, d_csr, d_cnt, j->d_jidx, jv);
    });
    if (r2 != HF_OK) return r2;
    return timed_launch("join_fixup", [&] {
      hipLaunchKernelGGL((k_join_fixup<NR>), dim3(2048), dim3(BLOCK), 0,
                         g.stream, d_csr, n_slots, j->d_jidx, jv, d_fix_err);
    });
  };
  switch (nr) {
#define HF_JB_CASE(NR) case NR: rc = fill(std::integral_constant<int, NR>{}); break;
    HF_JB_CASE(0) HF_JB_CASE(1) HF_JB_CASE(2) HF_JB_CASE(3) HF_JB_CASE(4)
    HF_JB_CASE(5) HF_JB_CASE(6) HF_JB_CASE(7) HF_JB_CASE(8)
#undef HF_JB_CASE
  }
  dev_free(d_cnt, g.stream);
  dev_free(d_tiles, g.stream);
  if (rc != HF_OK) { hf_join_free(j); return rc; }
  // surface hist/fixup errors
  unsigned long long h_err[2] = {0, 0};
  HF_HIP("hf_join_build",
         hipMemcpyAsync(&h_err[0], d_hist_err, 8, hipMemcpyDeviceToHost,
                        g.stream));
  HF_HIP("hf_join_build",
         hipMemcpyAsync(&h_err[1], d_fix_err, 8, hipMemcpyDeviceToHost,
                        g.stream));
  HF_HIP("hf_join_build", hipStreamSynchronize(g.stream));
  if (h_err[0]) {
    hipMemsetAsync(d_hist_err, 0, 8, g.stream);
    hipMemsetAsync(d_fix_err, 0, 8, g.stream);
    hf_join_free(j);
    char buf[160];
    snprintf(buf, sizeof buf,
             "%llu right keys outside [key_min, key_min+n_slots)",
             h_err[0]);
    return set_err(HF_ERR_ARG, "hf_join_build", buf);
  }
  if (h_err[1]) {
    // keys beyond the 4096-duplicate in-thread fixup (round 2): rebuild
    // ORDER-CORRECT from a stable ascending key sort — within equal keys
    // the stable sort preserves original right order, which is exactly
    // the CSR layout (the offsets are fill-independent and already
    // computed).  Removes the per-key multiplicity cap entirely.
    hipMemsetAsync(d_fix_err, 0, 8, g.stream);
    hf_col* perm = nullptr;
    int r3 = hf_sort_perm(rkeys, 1, &perm);
    if (r3 != HF_OK) { hf_join_free(j); return r3; }
    r3 = timed_launch("join_sortfill", [&] {
      hipLaunchKernelGGL(k_i64_to_u32, dim3((uint32_t)grid_for(n)),
                         dim3(BLOCK), 0, g.stream,
                         (const int64_t*)perm->dptr, j->d_jidx, n);
    });
    for (int c = 0; c < nr && r3 == HF_OK; ++c) {
      hf_col* sv = nullptr;
      r3 = hf_gather(rvals[c], perm, &sv);
      if (r3 != HF_OK) break;
      r3 = (hipMemcpyAsync(j->d_jval[c], sv->dptr, n * 8,
                           hipMemcpyDeviceToDevice, g.stream) == hipSuccess)
               ? HF_OK
               : set_err(HF_ERR_HIP, "hf_join_build", "sortfill copy");
      hf_col_free(sv);
    }
    hf_col_free(perm);
    if (r3 != HF_OK) { hf_join_free(j); return r3; }
  }
  *out = j;
  return HF_OK;
}

int hf_join_free(hf_join* j) {
  if (!j) return HF_OK;
  if (g.inited) {
    if (j->d_csr) dev_free(j->d_csr, g.stream);
    if (j->d_jidx) dev_free(j->d_jidx, g.stream);
    for (int c = 0; c < j->nr; ++c)
      if (j->d_jval[c]) dev_free(j->d_jval[c], g.stream);
  }
  delete j;
  return HF_OK;
}

int hf_join_probe(const hf_join* j, const hf_col* lkeys, hf_col** out_keys,
                  hf_col** out_lidx, hf_col** out_rcols, int64_t* n_out) {
  HF_NEED_INIT("hf_join_probe");
  if (!j || !lkeys || !out_keys || !out_lidx || !n_out)
    return set_err(HF_ERR_ARG, "hf_join_probe", "null");
  if (lkeys->dtype != HF_INT64)
    return set_err(HF_ERR_ARG, "hf_join_probe", "left keys must be int64");
  const int64_t n = lkeys->len;
  const int64_t ntiles = n > 0 ? (n + JOIN_TILE - 1) / JOIN_TILE : 1;
  unsigned long long* d_offs = nullptr;
  unsigned* d_cnts = nullptr;
  int64_t* d_tiles = nullptr;
  int64_t* d_total = (int64_t*)((char*)g.d_scratch + SCRATCH_NGROUPS);
  const int64_t alloc_n = n > 0 ? n : 1;
  HF_HIP("hf_join_probe",
         dev_alloc((void**)&d_offs, alloc_n * 8, g.stream));
  HF_HIP("hf_join_probe",
         dev_alloc((void**)&d_cnts, alloc_n * 4, g.stream));
  HF_HIP("hf_join_probe", dev_alloc((void**)&d_tiles, ntiles * 8, g.stream));
  HF_HIP("hf_join_probe", hipMemsetAsync(d_tiles, 0, ntiles * 8, g.stream));
  int rc = HF_OK;
  if (n > 0) {
    rc = timed_launch("join_probe_count", [&] {
      hipLaunchKernelGGL(k_probe_count, dim3((uint32_t)ntiles), dim3(BLOCK), 0,
                         g.stream, (const int64_t*)lkeys->dptr, n, j->key_min,
                         j->n_slots, j->d_csr, d_offs, d_cnts, d_tiles);
    });
    if (rc != HF_OK) return rc;
  }
  hipLaunchKernelGGL(k_compact_scan, dim3(1), dim3(1024), 0, g.stream, d_tiles,
                     ntiles, d_total);
  int64_t total = 0;
  HF_HIP("hf_join_probe",
         hipMemcpyAsync(&total, d_total, 8, hipMemcpyDeviceToHost, g.stream));
  HF_HIP("hf_join_probe", hipStreamSynchronize(g.stream));
  rc = hf_col_alloc(total, HF_INT64, out_keys);
  if (rc != HF_OK) return rc;
  rc = hf_col_alloc(total, HF_INT64, out_lidx);
  if (rc != HF_OK) return rc;
  JoinPtrs outr{};
  JoinConstPtrs jvc{};
  for (int c = 0; c < j->nr; ++c) {
    rc = hf_col_alloc(total, j->dtypes[c], &out_rcols[c]);
    if (rc != HF_OK) return rc;
    outr.vals[c] = (double*)out_rcols[c]->dptr;
    jvc.vals[c] = j->d_jval[c];
  }
  if (n > 0 && total > 0) {
    auto emit = [&](auto nrTag) {
      constexpr int NR = decltype(nrTag)::value;
      return timed_launch("join_probe_emit", [&] {
        hipLaunchKernelGGL((k_probe_emit<NR>), dim3((uint32_t)ntiles),
                           dim3(BLOCK), 0, g.stream,
                           (const int64_t*)lkeys->dptr, n, (int64_t)0, d_offs,
                           d_cnts, d_tiles, j->d_jidx, jvc,
                           (int64_t*)(*out_keys)->dptr,
                           (int64_t*)(*out_lidx)->dptr, outr);
      });
    };
    switch (j->nr) {
#define HF_JP_CASE(NR) case NR: rc = emit(std::integral_constant<int, NR>{}); break;
      HF_JP_CASE(0) HF_JP_CASE(1) HF_JP_CASE(2) HF_JP_CASE(3) HF_JP_CASE(4)
      HF_JP_CASE(5) HF_JP_CASE(6) HF_JP_CASE(7) HF_JP_CASE(8)
#undef HF_JP_CASE
    }
  }
  dev_free(d_offs, g.stream);
  dev_free(d_cnts, g.stream);
  dev_free(d_tiles, g.stream);
  if (rc != HF_OK) return rc;
  *n_out = total;
  return HF_OK;
}

int hf_gather(const hf_col* col, const hf_col* idx, hf_col** out) {
  HF_NEED_INIT("hf_gather");
  if (!col || !idx || !out) return set_err(HF_ERR_ARG, "hf_gather", "null");
  if (idx->dtype != HF_INT64)
    return set_err(HF_ERR_ARG, "hf_gather", "index must be int64");
  int rc = hf_col_alloc(idx->len, col->dtype, out);
  if (rc != HF_OK) return rc;
  const int64_t n = idx->len;
  if (n == 0) return HF_OK;
  rc = timed_launch("gather", [&] {
    if (col->dtype == HF_FLOAT64)
      hipLaunchKernelGGL(k_gather_f64, dim3((uint32_t)grid_for(n)), dim3(BLOCK),
                         0, g.stream, (const double*)col->dptr,
                         (const int64_t*)idx->dptr, (double*)(*out)->dptr, n);
    else
      hipLaunchKernelGGL(k_gather_i64, dim3((uint32_t)grid_for(n)), dim3(BLOCK),
                         0, g.stream, (const int64_t*)col->dptr,
                         (const int64_t*)idx->dptr, (int64_t*)(*out)->dptr, n);
  });
  if (rc != HF_OK) { hf_col_free(*out); *out = nullptr; }
  return rc;
}

int hf_sort_perm(const hf_col* keys, int ascending, hf_col** out_perm) {
  HF_NEED_INIT("hf_sort_perm");
  if (!keys || !out_perm) return set_err(HF_ERR_ARG, "hf_sort_perm", "null");
  if (keys->dtype != HF_INT64)
    return set_err(HF_ERR_ARG, "hf_sort_perm", "sort key must be int64");
  const int64_t n = keys->len;
  int rc = hf_col_alloc(n, HF_INT64, out_perm);
  if (rc != HF_OK) return rc;
  if (n == 0) return HF_OK;
  if (n >= (1LL << 31))
    return set_err(HF_ERR_UNSUPPORTED, "hf_sort_perm",
                   "partition too large (u32 origin indices)");
  hf_reduce_result r;
  rc = hf_reduce(keys, &r);
  if (rc != HF_OK) return rc;
  const int64_t key_min = r.imn;
  const uint64_t span = (uint64_t)r.imx - (uint64_t)r.imn;  // mod-2^64 safe
  if (span >= (1ULL << 27)) {
    // wide path: u64 shifted keys + u32 origins, up to 8 digit passes
    int bits = 0;
    while (bits < 64 && (span >> bits) != 0) ++bits;
    const int passes = (bits + 7) / 8;
    const int64_t ntiles = (n + SORT_TILE - 1) / SORT_TILE;
    const int64_t L = ntiles * 256;
    unsigned long long *kA = nullptr, *kB = nullptr, *offs = nullptr;
    unsigned *iA = nullptr, *iB = nullptr, *C = nullptr, *CT = nullptr;
    int64_t* scan_tiles = nullptr;
    const int64_t scan_tiles_n = (L + JOIN_TILE - 1) / JOIN_TILE;
    HF_HIP("hf_sort_perm", dev_alloc((void**)&kA, n * 8, g.stream));
    HF_HIP("hf_sort_perm", dev_alloc((void**)&kB, n * 8, g.stream));
    HF_HIP("hf_sort_perm", dev_alloc((void**)&iA, n * 4, g.stream));
    HF_HIP("hf_sort_perm", dev_alloc((void**)&iB, n * 4, g.stream));
    HF_HIP("hf_sort_perm", dev_alloc((void**)&C, L * 4, g.stream));
    HF_HIP("hf_sort_perm", dev_alloc((void**)&CT, L * 4, g.stream));
    HF_HIP("hf_sort_perm", dev_alloc((void**)&offs, (L + 1) * 8, g.stream));
    HF_HIP("hf_sort_perm",
           dev_alloc((void**)&scan_tiles, scan_tiles_n * 8, g.stream));
    int64_t* d_total = (int64_t*)((char*)g.d_scratch + SCRATCH_NGROUPS);
    rc = timed_launch("sort_pack", [&] {
      hipLaunchKernelGGL(k_sort_pack_wide, dim3((uint32_t)grid_for(n)),
                         dim3(BLOCK), 0, g.stream, (const int64_t*)keys->dptr,
                         n, key_min, span, ascending, kA, iA);
    });
    unsigned long long *kcur = kA, *kalt = kB;
    unsigned *icur = iA, *ialt = iB;
    const uint32_t wgrid =
        (uint32_t)std::min<int64_t>((ntiles + SORT_WPB - 1) / SORT_WPB, 2048);
    for (int p = 0; p < passes && rc == HF_OK; ++p) {
      const int shift = 8 * p;
      rc = timed_launch("sort_pass", [&] {
        hipLaunchKernelGGL(k_sort_count_wide, dim3(wgrid), dim3(BLOCK), 0,
                           g.stream, kcur, n, shift, C, ntiles);
        hipLaunchKernelGGL(k_transpose256,
                           dim3((uint32_t)((ntiles + 31) / 32), 8), dim3(1024),
                           0, g.stream, C, CT, ntiles);
        hipLaunchKernelGGL(k_tile_sums_u32, dim3((uint32_t)scan_tiles_n),
                           dim3(BLOCK), 0, g.stream, CT, L, scan_tiles);
        hipLaunchKernelGGL(k_compact_scan, dim3(1), dim3(1024), 0, g.stream,
                           scan_tiles, scan_tiles_n, d_total);
        hipLaunchKernelGGL(k_scan_apply_u32, dim3((uint32_t)scan_tiles_n),
                           dim3(BLOCK), 0, g.stream, CT, L, scan_tiles,
                           d_total, offs);
        hipLaunchKernelGGL(k_sort_scatter_wide, dim3(wgrid), dim3(BLOCK), 0,
                           g.stream, kcur, icur, n, shift, offs, ntiles, kalt,
                           ialt);
      });
      std::swap(kcur, kalt);
      std::swap(icur, ialt);
    }
    if (rc == HF_OK)
      rc = timed_launch("sort_unpack", [&] {
        hipLaunchKernelGGL(k_sort_unpack_wide, dim3((uint32_t)grid_for(n)),
                           dim3(BLOCK), 0, g.stream, icur, n,
                           (int64_t*)(*out_perm)->dptr);
      });
    dev_free(kA, g.stream);
    dev_free(kB, g.stream);
    dev_free(iA, g.stream);
    dev_free(iB, g.stream);
    dev_free(C, g.stream);
    dev_free(CT, g.stream);
    dev_free(offs, g.stream);
    dev_free(scan_tiles, g.stream);
    if (rc != HF_OK) { hf_col_free(*out_perm); *out_perm = nullptr; }
    return rc;
  }
  int bits = 0;
  while ((span >> bits) != 0) ++bits;
  const int passes = (bits + 7) / 8;
  const int64_t ntiles = (n + SORT_TILE - 1) / SORT_TILE;
  const int64_t L = ntiles * 256;
  unsigned long long *bufA = nullptr, *bufB = nullptr, *offs = nullptr;
  unsigned *C = nullptr, *CT = nullptr;
  int64_t* scan_tiles = nullptr;
  const int64_t scan_tiles_n = (L + JOIN_TILE - 1) / JOIN_TILE;
  HF_HIP("hf_sort_perm", dev_alloc((void**)&bufA, n * 8, g.stream));
  HF_HIP("hf_sort_perm", dev_alloc((void**)&bufB, n * 8, g.stream));
  HF_HIP("hf_sort_perm", dev_alloc((void**)&C, L * 4, g.stream));
  HF_HIP("hf_sort_perm", dev_alloc((void**)&CT, L * 4, g.stream));
  HF_HIP("hf_sort_perm", dev_alloc((void**)&offs, (L + 1) * 8, g.stream));
  HF_HIP("hf_sort_perm", dev_alloc((void**)&scan_tiles, scan_tiles_n * 8,
                                   g.stream));
  int64_t* d_total = (int64_t*)((char*)g.d_scratch + SCRATCH_NGROUPS);
  rc = timed_launch("sort_pack", [&] {
    hipLaunchKernelGGL(k_sort_pack, dim3((uint32_t)grid_for(n)), dim3(BLOCK),
                       0, g.stream, (const int64_t*)keys->dptr, n, key_min,
                       span, ascending, bufA);
  });
  unsigned long long* cur = bufA;
  unsigned long long* alt = bufB;
  const uint32_t wgrid =
      (uint32_t)std::min<int64_t>((ntiles + SORT_WPB - 1) / SORT_WPB, 2048);
  for (int p = 0; p < passes && rc == HF_OK; ++p) {
    const int shift = 8 * p;
    rc = timed_launch("sort_pass", [&] {
      hipLaunchKernelGGL(k_sort_count, dim3(wgrid), dim3(BLOCK), 0, g.stream,
                         cur, n, shift, C, ntiles);
      hipLaunchKernelGGL(k_transpose256,
                         dim3((uint32_t)((ntiles + 31) / 32), 8), dim3(1024),
                         0, g.stream, C, CT, ntiles);
      hipLaunchKernelGGL(k_tile_sums_u32, dim3((uint32_t)scan_tiles_n),
                         dim3(BLOCK), 0, g.stream, CT, L, scan_tiles);
      hipLaunchKernelGGL(k_compact_scan, dim3(1), dim3(1024), 0, g.stream,
                         scan_tiles, scan_tiles_n, d_total);
      hipLaunchKernelGGL(k_scan_apply_u32, dim3((uint32_t)scan_tiles_n),
                         dim3(BLOCK), 0, g.stream, CT, L, scan_tiles, d_total,
                         offs);
      hipLaunchKernelGGL(k_sort_scatter, dim3(wgrid), dim3(BLOCK), 0, g.stream,
                         cur, n, shift, offs, ntiles, alt);
    });
    std::swap(cur, alt);
  }
  if (rc == HF_OK)
    rc = timed_launch("sort_unpack", [&] {
      hipLaunchKernelGGL(k_sort_unpack, dim3((uint32_t)grid_for(n)),
                         dim3(BLOCK), 0, g.stream, cur,
                         n, (int64_t*)(*out_perm)->dptr);
    });
  dev_free(bufA, g.stream);
  dev_free(bufB, g.stream);
  dev_free(C, g.stream);
  dev_free(CT, g.stream);
  dev_free(offs, g.stream);
  dev_free(scan_tiles, g.stream);
  if (rc != HF_OK) { hf_col_free(*out_perm); *out_perm = nullptr; }
  return rc;
}

int hf_fill_f64(uintptr_t dptr, double value, int64_t n) {
  HF_NEED_INIT("hf_fill_f64");
  if (n <= 0) return HF_OK;
  return timed_launch("fill_f64", [&] {
    hipLaunchKernelGGL(k_fill_f64, dim3((uint32_t)grid_for(n)), dim3(BLOCK), 0,
                       g.stream, (double*)dptr, value, n);
  });
}

int hf_fixup_empty(const hf_col* val, const hf_col* cnt, hf_col** out) {
  HF_NEED_INIT("hf_fixup_empty");
  if (!val || !cnt || !out || val->len != cnt->len ||
      val->dtype != HF_FLOAT64 || cnt->dtype != HF_INT64)
    return set_err(HF_ERR_ARG, "hf_fixup_empty", "bad args");
  int rc = hf_col_alloc(val->len, HF_FLOAT64, out);
  if (rc != HF_OK) return rc;
  if (val->len == 0) return HF_OK;
  rc = timed_launch("fixup_empty", [&] {
    hipLaunchKernelGGL(k_fixup_empty, dim3((uint32_t)grid_for(val->len)),
                       dim3(BLOCK), 0, g.stream, (const double*)val->dptr,
                       (const int64_t*)cnt->dptr, (double*)(*out)->dptr,
                       val->len);
  });
  if (rc != HF_OK) { hf_col_free(*out); *out = nullptr; }
  return rc;
}

// ---- compare + filter ----

struct hf_filterplan {
  const long long* mask;   // borrowed: caller keeps the mask column alive
  int64_t n, total, ntiles;
  int64_t* d_tiles;        // exclusive per-tile kept offsets
};

static const int64_t* plan_tiles(const hf_filterplan* p) { return p->d_tiles; }

int hf_compare_scalar(int op, const hf_col* col, double scalar, hf_col** out) {
  HF_NEED_INIT("hf_compare_scalar");
  if (!col || !out) return set_err(HF_ERR_ARG, "hf_compare_scalar", "null");
  if (op < HF_CMP_GT || op > HF_CMP_NOTNA)
    return set_err(HF_ERR_ARG, "hf_compare_scalar", "unknown op");
  int rc = hf_col_alloc(col->len, HF_INT64, out);
  if (rc != HF_OK) return rc;
  const int64_t n = col->len;
  auto L = [&](auto opTag, auto tTag) {
    constexpr int O = decltype(opTag)::value;
    using T = typename decltype(tTag)::type;
    return timed_launch("compare", [&] {
      hipLaunchKernelGGL((k_compare<O, T>), dim3((uint32_t)grid_for(n)),
                         dim3(BLOCK), 0, g.stream, (const T*)col->dptr,
                         (long long*)(*out)->dptr, scalar, n);
    });
  };
  struct F64 { using type = double; };
  struct I64 { using type = long long; };
  const bool f = col->dtype == HF_FLOAT64;
  switch (op) {
#define HF_CMP_CASE(O)                                                         \
  case O:                                                                      \
    rc = f ? L(std::integral_constant<int, O>{}, F64{})                        \
           : L(std::integral_constant<int, O>{}, I64{});                       \
    break;
    HF_CMP_CASE(HF_CMP_GT) HF_CMP_CASE(HF_CMP_GE) HF_CMP_CASE(HF_CMP_LT)
    HF_CMP_CASE(HF_CMP_LE) HF_CMP_CASE(HF_CMP_EQ) HF_CMP_CASE(HF_CMP_NE)
    HF_CMP_CASE(HF_CMP_NOTNA)
#undef HF_CMP_CASE
  }
  if (rc != HF_OK) { hf_col_free(*out); *out = nullptr; }
  return rc;
}

int hf_filter_plan(const hf_col* mask, hf_filterplan** out, int64_t* n_kept) {
  HF_NEED_INIT("hf_filter_plan");
  if (!mask || !out || !n_kept)
    return set_err(HF_ERR_ARG, "hf_filter_plan", "null");
  if (mask->dtype != HF_INT64)
    return set_err(HF_ERR_ARG, "hf_filter_plan", "mask must be int64 0/1");
  const int64_t n = mask->len;
  const int64_t ntiles = n > 0 ? (n + FILT_TILE - 1) / FILT_TILE : 1;
  hf_filterplan* p = new hf_filterplan{};
  p->mask = (const long long*)mask->dptr;
  p->n = n;
  p->ntiles = ntiles;
  HF_HIP("hf_filter_plan",
         dev_alloc((void**)&p->d_tiles, ntiles * 8, g.stream));
  HF_HIP("hf_filter_plan", hipMemsetAsync(p->d_tiles, 0, ntiles * 8, g.stream));
  int64_t* d_total = (int64_t*)((char*)g.d_scratch + SCRATCH_NGROUPS);
  int rc = HF_OK;
  if (n > 0) {
    rc = timed_launch("filter_count", [&] {
      hipLaunchKernelGGL(k_filter_count, dim3((uint32_t)ntiles), dim3(BLOCK), 0,
                         g.stream, p->mask, n, p->d_tiles);
    });
    if (rc != HF_OK) { hf_filter_plan_free(p); return rc; }
  }
  hipLaunchKernelGGL(k_compact_scan, dim3(1), dim3(1024), 0, g.stream,
                     p->d_tiles, ntiles, d_total);
  HF_HIP("hf_filter_plan",
         hipMemcpyAsync(&p->total, d_total, 8, hipMemcpyDeviceToHost, g.stream));
  HF_HIP("hf_filter_plan", hipStreamSynchronize(g.stream));
  *n_kept = p->total;
  *out = p;
  return HF_OK;
}

int hf_filter_plan_free(hf_filterplan* p) {
  if (!p) return HF_OK;
  if (g.inited && p->d_tiles) dev_free(p->d_tiles, g.stream);
  delete p;
  return HF_OK;
}

int hf_filter_apply(const hf_filterplan* p, const hf_col* col, hf_col** out) {
  HF_NEED_INIT("hf_filter_apply");
  if (!p || !col || !out) return set_err(HF_ERR_ARG, "hf_filter_apply", "null");
  if (col->len != p->n)
    return set_err(HF_ERR_ARG, "hf_filter_apply", "length mismatch");
  int rc = hf_col_alloc(p->total, col->dtype, out);
  if (rc != HF_OK) return rc;
  if (p->n == 0 || p->total == 0) return HF_OK;
  rc = timed_launch("filter_scatter", [&] {
    if (col->dtype == HF_FLOAT64)
      hipLaunchKernelGGL((k_filter_scatter<double, false>),
                         dim3((uint32_t)p->ntiles), dim3(BLOCK), 0, g.stream,
                         p->mask, (const double*)col->dptr, p->n, p->d_tiles,
                         (int64_t)0, (double*)(*out)->dptr);
    else
      hipLaunchKernelGGL((k_filter_scatter<long long, false>),
                         dim3((uint32_t)p->ntiles), dim3(BLOCK), 0, g.stream,
                         p->mask, (const long long*)col->dptr, p->n, p->d_tiles,
                         (int64_t)0, (long long*)(*out)->dptr);
  });
  if (rc != HF_OK) { hf_col_free(*out); *out = nullptr; }
  return rc;
}

int hf_filter_iota(const hf_filterplan* p, int64_t base, hf_col** out) {
  HF_NEED_INIT("hf_filter_iota");
  if (!p || !out) return set_err(HF_ERR_ARG, "hf_filter_iota", "null");
  int rc = hf_col_alloc(p->total, HF_INT64, out);
  if (rc != HF_OK) return rc;
  if (p->n == 0 || p->total == 0) return HF_OK;
  rc = timed_launch("filter_iota", [&] {
    hipLaunchKernelGGL((k_filter_scatter<long long, true>),
                       dim3((uint32_t)p->ntiles), dim3(BLOCK), 0, g.stream,
                       p->mask, (const long long*)nullptr, p->n, p->d_tiles,
                       base, (long long*)(*out)->dptr);
  });
  if (rc != HF_OK) { hf_col_free(*out); *out = nullptr; }
  return rc;
}

// ---- profiling ----

int hf_profiling(int enable) {
  HF_NEED_INIT("hf_profiling");
  g.profiling = enable != 0;
  return HF_OK;
}

int hf_kernel_stats(const char* name, int64_t* launches, double* total_ms) {
  HF_NEED_INIT("hf_kernel_stats");
  int rc = resolve_stats("hf_kernel_stats");
  if (rc != HF_OK) return rc;
  auto it = g.stats.find(name);
  if (it == g.stats.end()) {
    *launches = 0;
    *total_ms = 0.0;
    return HF_OK;
  }
  *launches = it->second.first;
  *total_ms = it->second.second;
  return HF_OK;
}

int hf_kernel_stats_reset(void) {
  HF_NEED_INIT("hf_kernel_stats_reset");
  int rc = resolve_stats("hf_kernel_stats_reset");
  if (rc != HF_OK) return rc;
  g.stats.clear();
  return HF_OK;
}

}  // extern "C"
