// grapehip — MI355X (gfx950) GPU engine: kernels + host drivers.
//
// Hand-written CDNA4 HIP throughout (wave64, LDS-staged owner search, no
// Thrust/CUB/hipCUB). Reference parity map:
//   * cuda/parallel/parallel_engine.h LB schedulers -> expand_cm_* kernels
//     (LDS prefix owner-search; CTA/STRICT variants arrive in later rounds)
//   * cuda/utils/cuda_utils.h CUB scans -> hand-rolled hierarchical
//     exclusive scan (wave shfl scan + LDS cross-wave + recursive spine)
//   * cuda/parallel/gpu_message_manager.h -> HaloBuffer (global-vid dedup
//     bitmap + per-peer regions) exchanged with paired ncclSend/Recv over
//     xGMI, lengths via the TCP control plane
//   * cuda apps bfs/sssp/pagerank/wcc -> drivers below
#include "gpu_engine.hpp"

#include <rccl/rccl.h>

#include <algorithm>
#include <cmath>
#include <functional>
#include <cstring>
#include <unordered_map>

#include "dev_graph.hpp"

#define NCCL_CHECK(expr)                                                   \
  do {                                                                     \
    ncclResult_t _r = (expr);                                              \
    if (_r != ncclSuccess) {                                               \
      throw std::runtime_error(std::string("RCCL error: ") +               \
                               ncclGetErrorString(_r));                    \
    }                                                                      \
  } while (0)

namespace grapehip {

// ===========================================================================
// Device utilities
// ===========================================================================

__device__ __forceinline__ uint64_t mix64(uint64_t z) {
  z += 0x9e3779b97f4a7c15ULL;
  z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ULL;
  z = (z ^ (z >> 27)) * 0x94d049bb133111ebULL;
  return z ^ (z >> 31);
}

// atomicMin on non-negative float via ordered int reinterpret.
__device__ __forceinline__ float atomicMinPosFloat(float* addr, float val) {
  int old = atomicMin(reinterpret_cast<int*>(addr), __float_as_int(val));
  return __int_as_float(old);
}

// wave-inclusive scan (u64), 64 lanes
__device__ __forceinline__ uint64_t wave_incl_scan(uint64_t v) {
  const int lane = threadIdx.x & 63;
#pragma unroll
  for (int d = 1; d < 64; d <<= 1) {
    uint64_t up = __shfl_up(static_cast<unsigned long long>(v), d, 64);
    if (lane >= d) v += up;
  }
  return v;
}

// ===========================================================================
// Hierarchical exclusive scan: in (u32 or u64) -> out u64, returns total.
// Pass A: per-block sums; spine scanned recursively; pass B: final scan.
// ITEMS per thread keeps the spine small (405M rows -> ~200k blocks).
// ===========================================================================

constexpr int kScanItems = 8;
constexpr int kScanChunk = kBlock * kScanItems;  // 2048 elements per block

template <typename IN>
__global__ void scan_pass_a(const IN* __restrict__ in, size_t n,
                            uint64_t* __restrict__ bsums) {
  __shared__ uint64_t s_wave[kBlock / kWave];
  size_t base = static_cast<size_t>(blockIdx.x) * kScanChunk +
                static_cast<size_t>(threadIdx.x) * kScanItems;
  uint64_t sum = 0;
#pragma unroll
  for (int k = 0; k < kScanItems; ++k) {
    size_t i = base + k;
    if (i < n) sum += in[i];
  }
  // wave reduce
#pragma unroll
  for (int d = 32; d > 0; d >>= 1)
    sum += __shfl_down(static_cast<unsigned long long>(sum), d, 64);
  if ((threadIdx.x & 63) == 0) s_wave[threadIdx.x >> 6] = sum;
  __syncthreads();
  if (threadIdx.x == 0) {
    uint64_t t = 0;
#pragma unroll
    for (int w = 0; w < kBlock / kWave; ++w) t += s_wave[w];
    bsums[blockIdx.x] = t;
  }
}

template <typename IN>
__global__ void scan_pass_b(const IN* __restrict__ in, size_t n,
                            const uint64_t* __restrict__ boffs,
                            uint64_t* __restrict__ out) {
  __shared__ uint64_t s_wave[kBlock / kWave];
  size_t base = static_cast<size_t>(blockIdx.x) * kScanChunk +
                static_cast<size_t>(threadIdx.x) * kScanItems;
  uint64_t vals[kScanItems];
  uint64_t sum = 0;
#pragma unroll
  for (int k = 0; k < kScanItems; ++k) {
    size_t i = base + k;
    uint64_t v = i < n ? static_cast<uint64_t>(in[i]) : 0;
    vals[k] = sum;  // exclusive within thread
    sum += v;
  }
  uint64_t incl = wave_incl_scan(sum);
  uint64_t wave_excl = incl - sum;
  if ((threadIdx.x & 63) == 63) s_wave[threadIdx.x >> 6] = incl;
  __syncthreads();
  uint64_t wave_base = 0;
  for (int w = 0; w < static_cast<int>(threadIdx.x >> 6); ++w)
    wave_base += s_wave[w];
  uint64_t thread_base = boffs[blockIdx.x] + wave_base + wave_excl;
#pragma unroll
  for (int k = 0; k < kScanItems; ++k) {
    size_t i = base + k;
    if (i < n) out[i] = thread_base + vals[k];
  }
}

// Grow-only per-level scratch so per-round frontier scans never hipMalloc.
// Levels are FIXED-depth (2048^4 > 2^44 elements) so references returned by
// sum_buf/off_buf stay valid while deeper recursion levels allocate.
struct ScanTemp {
  static constexpr size_t kMaxLevels = 4;
  DeviceBuffer<uint64_t> sums[kMaxLevels];   // per level: block sums
  DeviceBuffer<uint64_t> offs[kMaxLevels];   // per level: scanned sums
  DeviceBuffer<uint64_t> zero;               // single 0 (spine base)
  DeviceBuffer<uint64_t>& sum_buf(size_t l, size_t n) {
    if (sums[l].size() < n) sums[l].resize(n + (n >> 2) + 16);
    return sums[l];
  }
  DeviceBuffer<uint64_t>& off_buf(size_t l, size_t n) {
    if (offs[l].size() < n) offs[l].resize(n + (n >> 2) + 16);
    return offs[l];
  }
};

template <typename IN>
void scan_recurse(const IN* d_in, uint64_t* d_out, size_t n, hipStream_t s,
                  ScanTemp& tmp, size_t level) {
  size_t nblocks = (n + kScanChunk - 1) / kScanChunk;
  if (nblocks <= 1) {
    if (tmp.zero.size() == 0) {
      tmp.zero.resize(1);
      tmp.zero.zero(s);
    }
    hipLaunchKernelGGL(scan_pass_b<IN>, dim3(1), dim3(kBlock), 0, s, d_in, n,
                       tmp.zero.data(), d_out);
    return;
  }
  if (level >= ScanTemp::kMaxLevels)
    throw std::runtime_error("scan: input too large");
  auto& bsums = tmp.sum_buf(level, nblocks);
  auto& boffs = tmp.off_buf(level, nblocks);
  hipLaunchKernelGGL(scan_pass_a<IN>, dim3(nblocks), dim3(kBlock), 0, s, d_in,
                     n, bsums.data());
  scan_recurse<uint64_t>(bsums.data(), boffs.data(), nblocks, s, tmp,
                         level + 1);
  hipLaunchKernelGGL(scan_pass_b<IN>, dim3(nblocks), dim3(kBlock), 0, s, d_in,
                     n, boffs.data(), d_out);
}

// Exclusive scan of n elements; d_out has n+1 slots (d_out[n] = total).
// Returns the total (host-synchronizing).
template <typename IN>
uint64_t exclusive_scan(const IN* d_in, uint64_t* d_out, size_t n,
                        hipStream_t s, ScanTemp& tmp) {
  if (n == 0) {
    uint64_t z = 0;
    HIP_CHECK(hipMemcpyAsync(d_out, &z, 8, hipMemcpyHostToDevice, s));
    HIP_CHECK(hipStreamSynchronize(s));
    return 0;
  }
  scan_recurse<IN>(d_in, d_out, n, s, tmp, 0);
  uint64_t last_off;
  IN last_in;
  HIP_CHECK(hipMemcpyAsync(&last_off, d_out + (n - 1), 8,
                           hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipMemcpyAsync(&last_in, d_in + (n - 1), sizeof(IN),
                           hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipStreamSynchronize(s));
  uint64_t total = last_off + static_cast<uint64_t>(last_in);
  HIP_CHECK(hipMemcpyAsync(d_out + n, &total, 8, hipMemcpyHostToDevice, s));
  HIP_CHECK(hipStreamSynchronize(s));
  return total;
}

}  // namespace grapehip

namespace grapehip {

// ===========================================================================
// Small device helpers: queues, bitmaps, fills
// ===========================================================================

struct DevQueue {
  uint32_t* q;
  unsigned long long* cnt;
  __device__ __forceinline__ void push(uint32_t v) const {
    q[atomicAdd(cnt, 1ull)] = v;
  }
};

struct DevBitmap {
  uint32_t* words;
  __device__ __forceinline__ bool set_once(uint32_t v) const {
    uint32_t m = 1u << (v & 31);
    return (atomicOr(&words[v >> 5], m) & m) == 0;
  }
  __device__ __forceinline__ void clear_bit(uint32_t v) const {
    atomicAnd(&words[v >> 5], ~(1u << (v & 31)));
  }
};

template <typename T>
__global__ void fill_kernel(T* p, T v, size_t n) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (size_t i = static_cast<size_t>(blockIdx.x) * blockDim.x + threadIdx.x;
       i < n; i += stride)
    p[i] = v;
}

template <typename T>
void fill(T* p, T v, size_t n, hipStream_t s) {
  if (n) fill_kernel<T><<<grid_for(n), kBlock, 0, s>>>(p, v, n);
}

__global__ void iota_kernel(uint32_t* p, uint32_t base, size_t n) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (size_t i = static_cast<size_t>(blockIdx.x) * blockDim.x + threadIdx.x;
       i < n; i += stride)
    p[i] = base + static_cast<uint32_t>(i);
}

// ===========================================================================
// Synthetic generation (RMAT) + CSR build
// ===========================================================================

// Quadrant thresholds in 16-bit fixed point.
// scr_a/scr_b: bijective LCG scramble (gcd(scr_a,nv)=1) decorrelating
// RMAT's low-popcount hub ids from the contiguous ownership slices —
// without it rank 0 of an 8-way run owns ~45% of all edges (measured),
// which would sink the scaling curve. scr_a=0 disables.
__global__ void gen_edges_kernel(uint64_t ne, uint64_t seed, int scale,
                                 uint32_t nv, uint32_t t_a, uint32_t t_ab,
                                 uint32_t t_abc, uint64_t scr_a,
                                 uint64_t scr_b, uint32_t my_begin,
                                 uint32_t my_end, bool undirected,
                                 bool reverse, bool weighted,
                                 uint32_t* out_src, uint32_t* out_dst,
                                 float* out_w,
                                 unsigned long long* out_cnt) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < ne; i += stride) {
    uint64_t h = mix64(seed ^ (i * 0x9e3779b97f4a7c15ULL));
    uint32_t s = 0, d = 0;
    int avail = 0;
    uint64_t bits = 0;
    for (int b = 0; b < scale; ++b) {
      if (avail == 0) {
        bits = h;
        h = mix64(h + 0x632be59bd9b4e019ULL);
        avail = 4;
      }
      uint32_t r = static_cast<uint32_t>(bits & 0xFFFF);
      bits >>= 16;
      --avail;
      uint32_t quad = r < t_a ? 0u : (r < t_ab ? 1u : (r < t_abc ? 2u : 3u));
      s = (s << 1) | (quad >> 1);
      d = (d << 1) | (quad & 1);
    }
    if (s >= nv) s -= nv;  // fold 2^scale domain onto [0, nv)
    if (d >= nv) d -= nv;
    if (scr_a) {
      // LCG then swap {0, scr_b}: raw vertex 0 (the deepest RMAT hub)
      // keeps id 0, so the benchmark's named source stays the hub while
      // every slice still gets a uniform share of the edge mass
      uint64_t ss = (scr_a * s + scr_b) % nv;
      uint64_t dd = (scr_a * d + scr_b) % nv;
      s = static_cast<uint32_t>(ss == scr_b ? 0 : (ss == 0 ? scr_b : ss));
      d = static_cast<uint32_t>(dd == scr_b ? 0 : (dd == 0 ? scr_b : dd));
    }
    float w = 1.0f;
    if (weighted)
      w = static_cast<float>((h >> 16) & 0xFFFFFF) * (99.0f / 16777216.0f) +
          1.0f;
    // materialize both orientations for undirected storage; keep owned src.
    // reverse=true keeps edges by OWNED DST as (d, s) pairs — the same
    // deterministic edge stream builds the in-CSR of a directed graph.
    // out_src==nullptr: counting pass only (RMAT mass is skewed toward low
    // vids, so per-rank shares are NOT ne/world — rank 0 of an 8-way run
    // owns far more; exact counts size the buffers)
    if (!reverse && s >= my_begin && s < my_end) {
      uint64_t pos = atomicAdd(out_cnt, 1ull);
      if (out_src) {
        out_src[pos] = s;
        out_dst[pos] = d;
        if (weighted) out_w[pos] = w;
      }
    }
    if (undirected && d >= my_begin && d < my_end) {
      // self loops stored twice, like the reference's undirected CSR
      uint64_t pos = atomicAdd(out_cnt, 1ull);
      if (out_src) {
        out_src[pos] = d;
        out_dst[pos] = s;
        if (weighted) out_w[pos] = w;
      }
    }
    if (reverse && d >= my_begin && d < my_end) {
      uint64_t pos = atomicAdd(out_cnt, 1ull);
      if (out_src) {
        out_src[pos] = d;
        out_dst[pos] = s;
        if (weighted) out_w[pos] = w;
      }
    }
  }
}

// ===========================================================================
// Hub-clustering renumber. RMAT/power-law hot vertices are the ids with
// few set bits — scattered across the id range, so the pull kernels'
// contrib/label gathers fetch a cold 64B line per 4B value. Renumbering
// each slice by descending (capped) degree packs the hot ~1M vertices
// into a contiguous L2-resident prefix (measured: top-1M ids carry ~53%
// of endpoint references at datagen-9_0 shape, but only 8% sit in any
// contiguous 1M window). Owner-preserving: new id = v_begin + position
// within the slice, so slicing, halos and collectives are untouched.
// ===========================================================================

constexpr int kRenumBuckets = 4096;  // degree-capped counting sort

__global__ void renum_hist_kernel(const uint32_t* __restrict__ deg,
                                  uint32_t n,
                                  unsigned long long* __restrict__ hist) {
  __shared__ uint32_t h[kRenumBuckets];
  for (int b = threadIdx.x; b < kRenumBuckets; b += blockDim.x) h[b] = 0;
  __syncthreads();
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint32_t k = deg[i];
    atomicAdd(&h[k < kRenumBuckets ? k : kRenumBuckets - 1], 1u);
  }
  __syncthreads();
  for (int b = threadIdx.x; b < kRenumBuckets; b += blockDim.x)
    if (h[b]) atomicAdd(&hist[b], static_cast<unsigned long long>(h[b]));
}

// block-synchronous 3-phase scatter: LDS-aggregated cursor claims keep the
// hot low-degree buckets off the global atomic units
__global__ void renum_scatter_kernel(const uint32_t* __restrict__ deg,
                                     uint32_t n, uint32_t v_begin,
                                     unsigned long long* __restrict__ cur,
                                     uint32_t* __restrict__ perm) {
  __shared__ uint32_t h[kRenumBuckets];
  __shared__ unsigned long long base[kRenumBuckets];
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t chunk = blockIdx.x * blockDim.x; chunk < n;
       chunk += stride) {
    uint32_t i = chunk + threadIdx.x;
    uint32_t k = kRenumBuckets;
    if (i < n) {
      uint32_t d = deg[i];
      k = d < kRenumBuckets ? d : kRenumBuckets - 1;
    }
    for (int b = threadIdx.x; b < kRenumBuckets; b += blockDim.x) h[b] = 0;
    __syncthreads();
    uint32_t mypos = 0;
    if (k < kRenumBuckets) mypos = atomicAdd(&h[k], 1u);
    __syncthreads();
    for (int b = threadIdx.x; b < kRenumBuckets; b += blockDim.x)
      if (h[b])
        base[b] = atomicAdd(&cur[b], static_cast<unsigned long long>(h[b]));
    __syncthreads();
    if (k < kRenumBuckets)
      perm[v_begin + i] =
          v_begin + static_cast<uint32_t>(base[k] + mypos);
    __syncthreads();
  }
}

__global__ void remap_ids_kernel(uint32_t* __restrict__ ids, uint64_t n,
                                 const uint32_t* __restrict__ perm) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < n; i += stride)
    ids[i] = perm[ids[i]];
}

__global__ void inv_scatter_kernel(const uint32_t* __restrict__ perm,
                                   uint32_t n, uint32_t* __restrict__ inv) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    inv[perm[i]] = i;
}

__global__ void count_deg_kernel(const uint32_t* src, uint64_t n,
                                 uint32_t v_begin, uint32_t* deg) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < n; i += stride)
    atomicAdd(&deg[src[i] - v_begin], 1u);
}

__global__ void scatter_edges_kernel(const uint32_t* src, const uint32_t* dst,
                                     const float* w, uint64_t n,
                                     uint32_t v_begin,
                                     unsigned long long* cursor,
                                     uint32_t* out_dst, float* out_w) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < n; i += stride) {
    uint64_t pos = atomicAdd(&cursor[src[i] - v_begin], 1ull);
    out_dst[pos] = dst[i];
    if (w) out_w[pos] = w[i];
  }
}

// ===========================================================================
// Segmented adjacency sort (ascending dst per CSR row). Applied once after
// the device CSR is built: ascending neighbor ids turn the pull kernels'
// random gathers into locally-increasing streams (L2-friendlier) and give
// deterministic adjacency order. Weighted rows sort (dst,weight) pairs
// packed into u64. LDS bitonic for rows <= 4096, pow2-padded global
// scratch bitonic for heavier rows.
// ===========================================================================

constexpr uint32_t kSegSortLds = 4096;

__device__ __forceinline__ void bitonic_stage_u64(unsigned long long* a,
                                                  uint32_t n, uint32_t k,
                                                  uint32_t j) {
  for (uint32_t i = threadIdx.x; i < n; i += blockDim.x) {
    uint32_t ij = i ^ j;
    if (ij > i) {
      bool up = (i & k) == 0;
      unsigned long long x = a[i], y = a[ij];
      if ((x > y) == up) {
        a[i] = y;
        a[ij] = x;
      }
    }
  }
}

__global__ void seg_sortbucket_kernel(const uint64_t* __restrict__ off,
                                      uint32_t rows_n, uint32_t* lds_rows,
                                      unsigned long long* c_lds,
                                      uint32_t* big_rows,
                                      unsigned long long* c_big) {
  __shared__ uint32_t s_cnt[2];
  __shared__ unsigned long long s_base[2];
  uint32_t* lists[2] = {lds_rows, big_rows};
  unsigned long long* gcnt[2] = {c_lds, c_big};
  const uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t base = blockIdx.x * blockDim.x; base < rows_n;
       base += stride) {
    if (threadIdx.x < 2) s_cnt[threadIdx.x] = 0;
    __syncthreads();
    uint32_t r = base + threadIdx.x;
    int b = -1;
    uint32_t loc = 0;
    if (r < rows_n) {
      uint64_t n = off[r + 1] - off[r];
      if (n >= 2) {
        b = n <= kSegSortLds ? 0 : 1;
        loc = atomicAdd(&s_cnt[b], 1u);
      }
    }
    __syncthreads();
    if (threadIdx.x < 2 && s_cnt[threadIdx.x])
      s_base[threadIdx.x] =
          atomicAdd(gcnt[threadIdx.x],
                    static_cast<unsigned long long>(s_cnt[threadIdx.x]));
    __syncthreads();
    if (b >= 0) lists[b][s_base[b] + loc] = r;
    __syncthreads();
  }
}

__global__ void seg_sort_lds_kernel(const uint64_t* __restrict__ off,
                                    uint32_t* __restrict__ dst,
                                    float* __restrict__ w,
                                    const uint32_t* __restrict__ rows,
                                    uint64_t nrows) {
  __shared__ unsigned long long s_a[kSegSortLds];
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t r = rows[i];
    uint64_t b = off[r];
    uint32_t n = static_cast<uint32_t>(off[r + 1] - b);
    uint32_t cap = 2;
    while (cap < n) cap <<= 1;
    for (uint32_t k = threadIdx.x; k < cap; k += blockDim.x)
      s_a[k] = k < n
                   ? ((static_cast<unsigned long long>(dst[b + k]) << 32) |
                      (w ? __float_as_uint(w[b + k]) : 0u))
                   : ~0ull;
    __syncthreads();
    for (uint32_t k = 2; k <= cap; k <<= 1)
      for (uint32_t j = k >> 1; j > 0; j >>= 1) {
        bitonic_stage_u64(s_a, cap, k, j);
        __syncthreads();
      }
    for (uint32_t k = threadIdx.x; k < n; k += blockDim.x) {
      dst[b + k] = static_cast<uint32_t>(s_a[k] >> 32);
      if (w) w[b + k] = __uint_as_float(static_cast<uint32_t>(s_a[k]));
    }
    __syncthreads();
  }
}

__global__ void seg_bigcap_kernel(const uint64_t* __restrict__ off,
                                  const uint32_t* __restrict__ rows,
                                  uint64_t nrows,
                                  uint32_t* __restrict__ caps) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < nrows; i += stride) {
    uint64_t n = off[rows[i] + 1] - off[rows[i]];
    uint64_t cap = 2;
    while (cap < n) cap <<= 1;
    caps[i] = static_cast<uint32_t>(cap);
  }
}

__global__ void seg_bigpad_kernel(const uint64_t* __restrict__ off,
                                  const uint32_t* __restrict__ dst,
                                  const float* __restrict__ w,
                                  const uint32_t* __restrict__ rows,
                                  uint64_t nrows,
                                  const uint64_t* __restrict__ pad_off,
                                  unsigned long long* __restrict__ scratch) {
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t r = rows[i];
    uint64_t b = off[r];
    uint64_t n = off[r + 1] - b;
    uint64_t pb = pad_off[i];
    uint64_t cap = pad_off[i + 1] - pb;
    for (uint64_t k = threadIdx.x; k < cap; k += blockDim.x)
      scratch[pb + k] =
          k < n ? ((static_cast<unsigned long long>(dst[b + k]) << 32) |
                   (w ? __float_as_uint(w[b + k]) : 0u))
                : ~0ull;
  }
}

__global__ void seg_bigsort_kernel(const uint64_t* __restrict__ pad_off,
                                   uint64_t nrows,
                                   unsigned long long* __restrict__ scratch) {
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint64_t pb = pad_off[i];
    uint32_t cap = static_cast<uint32_t>(pad_off[i + 1] - pb);
    unsigned long long* a = scratch + pb;
    for (uint32_t k = 2; k <= cap; k <<= 1)
      for (uint32_t j = k >> 1; j > 0; j >>= 1) {
        bitonic_stage_u64(a, cap, k, j);
        __syncthreads();
      }
  }
}

__global__ void seg_bigunpad_kernel(const uint64_t* __restrict__ off,
                                    uint32_t* __restrict__ dst,
                                    float* __restrict__ w,
                                    const uint32_t* __restrict__ rows,
                                    uint64_t nrows,
                                    const uint64_t* __restrict__ pad_off,
                                    const unsigned long long* __restrict__
                                        scratch) {
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t r = rows[i];
    uint64_t b = off[r];
    uint64_t n = off[r + 1] - b;
    uint64_t pb = pad_off[i];
    for (uint64_t k = threadIdx.x; k < n; k += blockDim.x) {
      dst[b + k] = static_cast<uint32_t>(scratch[pb + k] >> 32);
      if (w) w[b + k] = __uint_as_float(
                 static_cast<uint32_t>(scratch[pb + k]));
    }
  }
}

// ===========================================================================
// CM-style load-balanced edge expansion (LDS-staged owner search).
// Each block claims 256 consecutive work items (frontier entries or rows),
// stages their frontier-edge prefix + adjacency base in LDS, then its 256
// threads sweep the chunk's edges, binary-searching the owner in LDS.
// Reference semantics: cuda/parallel/parallel_engine.h LBCM (:718-771).
// ===========================================================================

template <typename EdgeOp>
__global__ void expand_cm_frontier(DevGraphView g, const uint32_t* __restrict__ q,
                                   uint32_t qn, const uint64_t* __restrict__ foff,
                                   EdgeOp op) {
  __shared__ uint64_t s_off[kBlock + 1];   // frontier-space edge prefix
  __shared__ uint64_t s_base[kBlock];      // CSR base of each vertex
  __shared__ uint32_t s_v[kBlock];
  const int tid = threadIdx.x;
  for (uint32_t chunk = blockIdx.x * kBlock; chunk < qn;
       chunk += gridDim.x * kBlock) {
    const int n = min(static_cast<uint32_t>(kBlock), qn - chunk);
    if (tid < n) s_off[tid] = foff[chunk + tid];
    if (tid == 0) s_off[n] = foff[chunk + n];
    if (tid < n) {
      uint32_t v = q[chunk + tid];
      s_v[tid] = v;
      s_base[tid] = g.oe_off[g.row(v)];
    }
    __syncthreads();
    const uint64_t e0 = s_off[0], e1 = s_off[n];
    // a thread's edges are strided, so its owner index only moves forward:
    // binary-search once, then advance incrementally (amortized O(1) LDS)
    int lo = 0;
    {
      uint64_t e = e0 + tid;
      if (e < e1) {
        int hi = n - 1;
        while (lo < hi) {
          int mid = (lo + hi + 1) >> 1;
          if (s_off[mid] <= e) lo = mid; else hi = mid - 1;
        }
      }
    }
    for (uint64_t e = e0 + tid; e < e1; e += blockDim.x) {
      while (lo + 1 < n && s_off[lo + 1] <= e) ++lo;
      const uint32_t u = s_v[lo];
      const uint64_t eid = s_base[lo] + (e - s_off[lo]);
      op(u, g.oe_dst[eid], g.oe_w ? g.oe_w[eid] : 1.0f);
    }
    __syncthreads();
  }
}

// Whole-owned-range variant (PR/WCC): work item i == row i; the global CSR
// offsets ARE the prefix — no frontier scan needed.
template <bool USE_IN, typename EdgeOp>
__global__ void expand_cm_range(DevGraphView g, EdgeOp op) {
  __shared__ uint64_t s_off[kBlock + 1];
  const int tid = threadIdx.x;
  const uint32_t rows = g.owned();
  const uint64_t* off = USE_IN ? g.ie_off : g.oe_off;
  const uint32_t* dst = USE_IN ? g.ie_dst : g.oe_dst;
  const float* wt = USE_IN ? g.ie_w : g.oe_w;
  for (uint32_t chunk = blockIdx.x * kBlock; chunk < rows;
       chunk += gridDim.x * kBlock) {
    const int n = min(static_cast<uint32_t>(kBlock), rows - chunk);
    if (tid < n) s_off[tid] = off[chunk + tid];
    if (tid == 0) s_off[n] = off[chunk + n];
    __syncthreads();
    const uint64_t e0 = s_off[0], e1 = s_off[n];
    int lo = 0;
    {
      uint64_t e = e0 + tid;
      if (e < e1) {
        int hi = n - 1;
        while (lo < hi) {
          int mid = (lo + hi + 1) >> 1;
          if (s_off[mid] <= e) lo = mid; else hi = mid - 1;
        }
      }
    }
    for (uint64_t e = e0 + tid; e < e1; e += blockDim.x) {
      while (lo + 1 < n && s_off[lo + 1] <= e) ++lo;
      op(g.v_begin + chunk + lo, dst[e], wt ? wt[e] : 1.0f);
    }
    __syncthreads();
  }
}

// STRICT load balancing (reference cuda/parallel/parallel_engine.h
// LBSTRICT :883-979): perfectly edge-balanced — every block owns exactly
// ceil(total/grid) edge outputs regardless of row skew. Row ownership is
// found by binary search in the frontier edge prefix; the search is
// re-anchored once per thread and advanced incrementally (a thread's
// edges are strided, so its owner only moves forward).
__device__ __forceinline__ uint32_t strict_owner(
    const uint64_t* __restrict__ foff, uint32_t qn, uint64_t e) {
  uint32_t lo = 0, hi = qn;  // find last row with foff[row] <= e
  while (lo + 1 < hi) {
    uint32_t mid = (lo + hi) >> 1;
    if (foff[mid] <= e)
      lo = mid;
    else
      hi = mid;
  }
  return lo;
}

template <typename EdgeOp>
__global__ void expand_strict_frontier(DevGraphView g,
                                       const uint32_t* __restrict__ q,
                                       uint32_t qn,
                                       const uint64_t* __restrict__ foff,
                                       uint64_t total, EdgeOp op) {
  uint64_t per = (total + gridDim.x - 1) / gridDim.x;
  uint64_t eb = per * blockIdx.x;
  if (eb >= total) return;
  uint64_t ee = eb + per < total ? eb + per : total;
  uint64_t e = eb + threadIdx.x;
  if (e >= ee) return;
  uint32_t owner = strict_owner(foff, qn, e);
  for (; e < ee; e += blockDim.x) {
    while (owner + 1 < qn && foff[owner + 1] <= e) ++owner;
    uint32_t u = q[owner];
    uint64_t eid = g.oe_off[g.row(u)] + (e - foff[owner]);
    op(u, g.oe_dst[eid], g.oe_w ? g.oe_w[eid] : 1.0f);
  }
}

// WM: wave-granular CM (reference LBWARP, parallel_engine.h:775-845,
// re-derived for 64-wide waves). Each wave claims 64 frontier entries and
// stages their adjacency bases + degree prefix in REGISTERS (no LDS):
// the prefix comes from a wave inclusive scan, owner search probes
// lane-held prefix values with __shfl (6-step binary search per edge).
template <typename EdgeOp>
__global__ void expand_wm_frontier(DevGraphView g,
                                   const uint32_t* __restrict__ q,
                                   uint32_t qn, EdgeOp op) {
  const int lane = threadIdx.x & 63;
  const uint64_t wid =
      (static_cast<uint64_t>(blockIdx.x) * blockDim.x + threadIdx.x) /
      kWave;
  const uint64_t waves =
      (static_cast<uint64_t>(gridDim.x) * blockDim.x) / kWave;
  for (uint64_t c = wid * kWave; c < qn; c += waves * kWave) {
    uint32_t i = static_cast<uint32_t>(c) + lane;
    bool act = i < qn;
    uint32_t u = act ? q[i] : 0;
    uint64_t b = 0;
    uint32_t deg = 0;
    if (act) {
      uint32_t r = g.row(u);
      b = g.oe_off[r];
      deg = static_cast<uint32_t>(g.oe_off[r + 1] - b);
    }
    uint64_t incl = wave_incl_scan(deg);
    uint64_t total = __shfl(static_cast<unsigned long long>(incl), 63, 64);
    uint64_t excl = incl - deg;
    for (uint64_t e = lane; e < total; e += kWave) {
      int lo = 0, hi = 63;
#pragma unroll
      for (int step = 0; step < 6; ++step) {
        int mid = (lo + hi + 1) >> 1;
        uint64_t px =
            __shfl(static_cast<unsigned long long>(excl), mid, 64);
        if (px <= e) lo = mid;
        else hi = mid - 1;
      }
      uint32_t uo = __shfl(u, lo, 64);
      uint64_t bo = __shfl(static_cast<unsigned long long>(b), lo, 64);
      uint64_t po = __shfl(static_cast<unsigned long long>(excl), lo, 64);
      uint64_t eid = bo + (e - po);
      op(uo, g.oe_dst[eid], g.oe_w ? g.oe_w[eid] : 1.0f);
    }
  }
}

// Thread-per-item baseline (LB=none), frontier form.
template <typename EdgeOp>
__global__ void expand_none_frontier(DevGraphView g,
                                     const uint32_t* __restrict__ q,
                                     uint32_t qn, EdgeOp op) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < qn;
       i += stride) {
    uint32_t u = q[i];
    uint32_t r = g.row(u);
    uint64_t b = g.oe_off[r], e = g.oe_off[r + 1];
    for (uint64_t k = b; k < e; ++k)
      op(u, g.oe_dst[k], g.oe_w ? g.oe_w[k] : 1.0f);
  }
}

__global__ void gather_deg_kernel(DevGraphView g, const uint32_t* q,
                                  uint32_t qn, uint32_t* deg) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < qn;
       i += stride) {
    uint32_t r = g.row(q[i]);
    deg[i] = static_cast<uint32_t>(g.oe_off[r + 1] - g.oe_off[r]);
  }
}

}  // namespace grapehip

namespace grapehip {

// ===========================================================================
// Halo exchange (the GPUMessageManager, MI355X-shaped).
// Appends are deduped through a global-vid bitmap; per-peer index regions are
// packed into (v, state[v]) pairs at flush and exchanged with paired
// ncclSend/ncclRecv over xGMI; lengths travel on the TCP control plane.
// ===========================================================================

template <typename T>
struct HaloPair {
  uint32_t v;
  T val;
};

struct DevHalo {
  uint32_t* idx;                 // [world * cap]
  unsigned long long* cnt;       // [world]
  DevBitmap bm;                  // global-vid dedup
  uint64_t cap;
  uint32_t slice;
  int world;
  __device__ __forceinline__ void add(uint32_t v) const {
    if (bm.set_once(v)) {
      uint32_t o = v / slice;
      if (static_cast<int>(o) >= world) o = world - 1;
      idx[o * cap + atomicAdd(&cnt[o], 1ull)] = v;
    }
  }
};

template <typename T>
__global__ void halo_pack_kernel(const uint32_t* __restrict__ idx, uint64_t n,
                                 const T* __restrict__ state, DevBitmap bm,
                                 HaloPair<T>* __restrict__ out) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < n; i += stride) {
    uint32_t v = idx[i];
    out[i].v = v;
    out[i].val = state[v];
    bm.clear_bit(v);
  }
}

template <typename T, typename Op>
__global__ void halo_process_kernel(const HaloPair<T>* __restrict__ pairs,
                                    uint64_t n, Op op) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < n; i += stride)
    op(pairs[i].v, pairs[i].val);
}

__global__ void clear_bits_kernel(const uint32_t* __restrict__ q, uint32_t n,
                                  DevBitmap bm) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    bm.clear_bit(q[i]);
}

// ===========================================================================
// Bitmap frontier compaction: mark bits during expansion (1 atomicOr per
// improvement instead of bitmap+queue atomics), then popc+scan+fill to
// materialize the queue. Clears the bitmap as it fills.
// ===========================================================================

__global__ void popc_words_kernel(const uint32_t* __restrict__ words,
                                  size_t nwords, uint32_t* __restrict__ out) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (size_t i = static_cast<size_t>(blockIdx.x) * blockDim.x + threadIdx.x;
       i < nwords; i += stride)
    out[i] = __popc(words[i]);
}

__global__ void fill_frontier_kernel(uint32_t* __restrict__ words,
                                     size_t nwords,
                                     const uint64_t* __restrict__ woff,
                                     uint32_t bit_base,
                                     uint32_t* __restrict__ q) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (size_t i = static_cast<size_t>(blockIdx.x) * blockDim.x + threadIdx.x;
       i < nwords; i += stride) {
    uint32_t bits = words[i];
    if (!bits) continue;
    uint64_t base = woff[i];
    while (bits) {
      int b = __builtin_ctz(bits);
      bits &= bits - 1;
      q[base++] = bit_base + (static_cast<uint32_t>(i) << 5) + b;
    }
    words[i] = 0;
  }
}

// ===========================================================================
// App edge-op functors
// ===========================================================================

struct BfsOp {
  uint32_t* depth;
  uint32_t next_depth;
  DevBitmap next_bm;  // over owned bits
  uint32_t v_begin, v_end;
  bool multi;
  DevHalo halo;
  __device__ __forceinline__ void operator()(uint32_t, uint32_t d,
                                             float) const {
    // cheap cached read first: most edges hit settled vertices
    if (depth[d] <= next_depth) return;
    if (atomicMin(&depth[d], next_depth) > next_depth) {
      if (d >= v_begin && d < v_end)
        next_bm.set_once(d - v_begin);
      else if (multi)
        halo.add(d);
    }
  }
};

struct BfsRecvOp {  // incoming (v, depth) — v owned here
  uint32_t* depth;
  DevBitmap next_bm;
  uint32_t v_begin;
  __device__ __forceinline__ void operator()(uint32_t v, uint32_t dv) const {
    if (atomicMin(&depth[v], dv) > dv) next_bm.set_once(v - v_begin);
  }
};

__global__ void bfs_seed_kernel(uint32_t* depth, uint32_t src,
                                uint32_t v_begin, uint32_t* bm_words) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    depth[src] = 0;
    uint32_t r = src - v_begin;
    bm_words[r >> 5] |= 1u << (r & 31);
  }
}

struct SsspOp {
  float* dist;
  float prio_hi;
  DevBitmap near_bm;  // over owned bits; compacted to a queue per round
  DevQueue far_q;
  DevBitmap far_bm;  // over global vids (dedup across rounds)
  uint32_t v_begin, v_end;
  bool multi;
  DevHalo halo;
  __device__ __forceinline__ void bucket(uint32_t d, float nd) const {
    if (d >= v_begin && d < v_end) {
      if (nd < prio_hi) {
        near_bm.set_once(d - v_begin);
      } else {
        if (far_bm.set_once(d)) far_q.push(d);
      }
    } else if (multi) {
      halo.add(d);
    }
  }
  __device__ __forceinline__ void operator()(uint32_t u, uint32_t d,
                                             float w) const {
    float nd = dist[u] + w;
    if (dist[d] <= nd) return;  // cached read gate before the atomic
    float old = atomicMinPosFloat(&dist[d], nd);
    if (nd < old) bucket(d, nd);
  }
};

struct SsspRecvOp {
  float* dist;
  float prio_hi;
  DevBitmap near_bm;
  DevQueue far_q;
  DevBitmap far_bm;
  uint32_t v_begin;
  __device__ __forceinline__ void operator()(uint32_t v, float dv) const {
    float old = atomicMinPosFloat(&dist[v], dv);
    if (dv < old) {
      if (dv < prio_hi) {
        near_bm.set_once(v - v_begin);
      } else {
        if (far_bm.set_once(v)) far_q.push(v);
      }
    }
  }
};

__global__ void sssp_seed_kernel(float* dist, uint32_t src, uint32_t v_begin,
                                 uint32_t* near_words) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    dist[src] = 0.0f;
    uint32_t r = src - v_begin;
    near_words[r >> 5] |= 1u << (r & 31);
  }
}

// far-queue repartition after a priority advance
__global__ void sssp_repart_kernel(const uint32_t* __restrict__ far_in,
                                   uint32_t n, const float* __restrict__ dist,
                                   float prio_hi, DevBitmap near_bm,
                                   uint32_t v_begin, DevQueue far_out,
                                   DevBitmap far_bm) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    uint32_t v = far_in[i];
    if (dist[v] < prio_hi) {
      far_bm.clear_bit(v);
      near_bm.set_once(v - v_begin);
    } else {
      far_out.push(v);  // keeps its far bit
    }
  }
}

struct PrPushOp {
  const float* __restrict__ contrib;  // per owned row
  double* acc;
  uint32_t v_begin;
  __device__ __forceinline__ void operator()(uint32_t u, uint32_t d,
                                             float) const {
    unsafeAtomicAdd(&acc[d], static_cast<double>(contrib[u - v_begin]));
  }
};

// fp32 contributions (the reference's GPU PageRank precision,
// cuda/pagerank/pagerank.h:27-35 rank_t=float): halves the random-gather
// bytes of the pull pass; the rank/accumulator side stays fp64.
__global__ void pr_contrib_kernel(const double* __restrict__ rank,
                                  const uint64_t* __restrict__ off,
                                  uint32_t owned, uint32_t v_begin,
                                  float* __restrict__ contrib) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < owned;
       r += stride) {
    uint64_t deg = off[r + 1] - off[r];
    contrib[r] = deg ? static_cast<float>(rank[v_begin + r] /
                                          static_cast<double>(deg))
                     : 0.0f;
  }
}

__global__ void pr_apply_kernel(double* __restrict__ rank,
                                const double* __restrict__ acc,
                                const double* __restrict__ dangling,
                                double inv_n, double damping,
                                uint32_t v_begin, uint32_t owned,
                                double* __restrict__ l1_delta) {
  __shared__ double s_wave[kBlock / kWave];
  // base recomputed on device so the whole iteration is hipGraph-capturable
  const double base =
      (1.0 - damping) * inv_n + damping * (*dangling) * inv_n;
  double d1 = 0;
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < owned;
       r += stride) {
    double nv = base + damping * acc[v_begin + r];
    if (l1_delta) d1 += fabs(nv - rank[v_begin + r]);
    rank[v_begin + r] = nv;
  }
  if (!l1_delta) return;
#pragma unroll
  for (int d = 32; d > 0; d >>= 1) d1 += __shfl_down(d1, d, 64);
  if ((threadIdx.x & 63) == 0) s_wave[threadIdx.x >> 6] = d1;
  __syncthreads();
  if (threadIdx.x == 0) {
    double t = 0;
#pragma unroll
    for (int w = 0; w < kBlock / kWave; ++w) t += s_wave[w];
    unsafeAtomicAdd(l1_delta, t);
  }
}

// Fused apply + NEXT iteration's contrib/dangling: one pass over rank
// and the out-degree offsets instead of three (apply, then contrib, then
// dangling each re-reading rank). Single-GPU pull path; rows beyond
// owned_real (hashmap padding) get the rank update only.
__global__ void pr_fused_apply_kernel(
    double* __restrict__ rank, const double* __restrict__ acc,
    const double* __restrict__ dangling_cur, double inv_n, double damping,
    uint32_t v_begin, uint32_t owned, uint32_t owned_real,
    const uint64_t* __restrict__ off, float* __restrict__ contrib,
    double* __restrict__ dangling_next, double* __restrict__ l1_delta) {
  __shared__ double s_wave[kBlock / kWave];
  __shared__ double s_dang[kBlock / kWave];
  const double base =
      (1.0 - damping) * inv_n + damping * (*dangling_cur) * inv_n;
  double d1 = 0, dang = 0;
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < owned;
       r += stride) {
    double nv = base + damping * acc[v_begin + r];
    if (l1_delta) d1 += fabs(nv - rank[v_begin + r]);
    rank[v_begin + r] = nv;
    if (r < owned_real) {
      uint64_t deg = off[r + 1] - off[r];
      if (deg) {
        contrib[r] = static_cast<float>(nv / static_cast<double>(deg));
      } else {
        contrib[r] = 0.0f;
        dang += nv;
      }
    }
  }
#pragma unroll
  for (int d = 32; d > 0; d >>= 1) {
    dang += __shfl_down(dang, d, 64);
    if (l1_delta) d1 += __shfl_down(d1, d, 64);
  }
  if ((threadIdx.x & 63) == 0) {
    s_dang[threadIdx.x >> 6] = dang;
    s_wave[threadIdx.x >> 6] = d1;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    double td = 0, t1 = 0;
#pragma unroll
    for (int w = 0; w < kBlock / kWave; ++w) {
      td += s_dang[w];
      t1 += s_wave[w];
    }
    if (td) unsafeAtomicAdd(dangling_next, td);
    if (l1_delta && t1) unsafeAtomicAdd(l1_delta, t1);
  }
}

__global__ void pr_dangling_kernel(const double* __restrict__ rank,
                                   const uint64_t* __restrict__ off,
                                   uint32_t owned, uint32_t v_begin,
                                   double* __restrict__ out) {
  __shared__ double s_wave[kBlock / kWave];
  double sum = 0;
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < owned;
       r += stride)
    if (off[r + 1] == off[r]) sum += rank[v_begin + r];
#pragma unroll
  for (int d = 32; d > 0; d >>= 1) sum += __shfl_down(sum, d, 64);
  if ((threadIdx.x & 63) == 0) s_wave[threadIdx.x >> 6] = sum;
  __syncthreads();
  if (threadIdx.x == 0) {
    double t = 0;
#pragma unroll
    for (int w = 0; w < kBlock / kWave; ++w) t += s_wave[w];
    unsafeAtomicAdd(out, t);
  }
}

// WCC: min-root union-find with path halving (Afforest-style hooks).
__device__ __forceinline__ uint32_t wcc_find(uint32_t* parent, uint32_t v) {
  for (;;) {
    uint32_t p = parent[v];
    if (p == v) return v;
    uint32_t gp = parent[p];
    if (gp == p) return p;
    // path halving MUST be monotone: a plain store can overwrite a
    // concurrently-lowered entry with a stale-derived higher value — or
    // even re-self-root it (gp read stale == v), severing a hook link
    // and splitting the component (seen at 1M-vertex scale)
    atomicMin(&parent[v], gp);
    v = gp;
  }
}

struct WccOp {
  uint32_t* parent;
  int* changed;
  uint32_t* mark_words;  // multi-GPU: entries hooked this round (for the
                         // sparse pair merge); nullptr on one GPU
  __device__ __forceinline__ void operator()(uint32_t u, uint32_t v,
                                             float) const {
    for (;;) {
      uint32_t ru = wcc_find(parent, u);
      uint32_t rv = wcc_find(parent, v);
      if (ru == rv) return;
      uint32_t hi = ru > rv ? ru : rv, lo = ru > rv ? rv : ru;
      uint32_t old = atomicCAS(&parent[hi], hi, lo);
      if (old == hi) {
        *changed = 1;
        if (mark_words)
          atomicOr(&mark_words[hi >> 5], 1u << (hi & 31));
        return;
      }
      // CAS returned the CURRENT entry — chase from it rather than
      // re-reading parent[hi] (a stale cached line would spin forever)
      u = old;
      v = lo;
    }
  }
};

__global__ void wcc_compress_kernel(uint32_t* parent, uint32_t n,
                                    int* changed = nullptr) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t v = blockIdx.x * blockDim.x + threadIdx.x; v < n;
       v += stride) {
    uint32_t r = wcc_find(parent, v);
    if (parent[v] != r) {
      atomicMin(&parent[v], r);  // monotone (see wcc_find)
      if (changed) *changed = 1;
    }
  }
}

// Output canonicalization. wcc_find's path-halving store races
// non-monotonically (a stale-derived higher value can overwrite a
// concurrently lowered entry), so a "no-change" wcc_compress pass can
// still leave entries pointing at non-roots. This variant chases
// READ-ONLY and writes only true static-forest roots — two clean passes
// guarantee parent[v] == root(v) exactly.
__global__ void wcc_finalize_kernel(uint32_t* __restrict__ parent,
                                    uint32_t n, int* __restrict__ changed) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t v = blockIdx.x * blockDim.x + threadIdx.x; v < n;
       v += stride) {
    uint32_t r = v;
    uint32_t p;
    while ((p = parent[r]) != r) r = p;
    if (parent[v] != r) {
      parent[v] = r;
      *changed = 1;
    }
  }
}

// pack (v, parent[v]) pairs for the sparse multi-GPU merge
__global__ void wcc_pack_pairs_kernel(const uint32_t* __restrict__ q,
                                      uint64_t n,
                                      const uint32_t* __restrict__ parent,
                                      uint32_t* __restrict__ out) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < n; i += stride) {
    uint32_t v = q[i];
    out[2 * i] = v;
    out[2 * i + 1] = parent[v];
  }
}

__global__ void wcc_apply_pairs_kernel(const uint32_t* __restrict__ pairs,
                                       uint64_t n, uint64_t skip_b,
                                       uint64_t skip_e,
                                       uint32_t* __restrict__ parent) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < n; i += stride) {
    if (i >= skip_b && i < skip_e) continue;  // own pairs already applied
    atomicMin(&parent[pairs[2 * i]], pairs[2 * i + 1]);
  }
}

// Afforest-style neighbor sampling (Sutton et al.): hook only the first K
// edges of each row — a cheap streaming pass that already collapses most
// of a power-law graph into its giant component.
__global__ void wcc_sample_kernel(DevGraphView g, int which,
                                  uint32_t* parent, int* changed,
                                  uint32_t* mark_words) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  const uint32_t rows = g.owned();
  WccOp op{parent, changed, mark_words};
  for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < rows;
       r += stride) {
    uint64_t b = g.oe_off[r], e = g.oe_off[r + 1];
    uint64_t i = b + which;
    if (i < e) op(g.v_begin + r, g.oe_dst[i], 1.0f);
  }
}

// mark owned rows NOT in the giant component (bitmap -> frontier queue);
// their remaining edges are the only ones the full pass must visit
__global__ void wcc_mark_rest_kernel(const uint32_t* __restrict__ parent,
                                     const uint64_t* __restrict__ off,
                                     uint32_t owned, uint32_t v_begin,
                                     uint32_t giant, DevBitmap bm) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < owned;
       r += stride)
    if (parent[v_begin + r] != giant && off[r + 1] != off[r])
      bm.set_once(r);
}

}  // namespace grapehip

namespace grapehip {

// ===========================================================================
// Degree-bucketed row scheduler + pull-PageRank kernels.
// Pull (gather) PageRank has NO atomics: with undirected symmetric storage
// (or the in-CSR for directed graphs) every owned row's full neighborhood is
// local, so each row's accumulator is a private sum — the fp64-atomic push
// was 10x slower at datagen-9_0 scale. Rows are bucketed by degree once:
// thread-per-row (<64), wave-per-row (<16384), block-per-row (rest) — the
// preprocessed analogue of the reference's CTA scheduler
// (cuda/parallel/parallel_engine.h LBCTA :849-879).
// ===========================================================================

constexpr uint64_t kSmallDeg = 64;
constexpr uint64_t kMidDeg = 16384;

// Block-aggregated: LDS tallies + one global reservation per block per
// bucket (405M per-thread global atomics on 3 hot words took 4.8 s).
__global__ void bucket_rows_kernel(const uint64_t* __restrict__ off,
                                   uint32_t owned, uint32_t* sm,
                                   unsigned long long* cs, uint32_t* md,
                                   unsigned long long* cm, uint32_t* lg,
                                   unsigned long long* cl) {
  __shared__ uint32_t s_cnt[3];
  __shared__ unsigned long long s_base[3];
  uint32_t* lists[3] = {sm, md, lg};
  unsigned long long* gcnt[3] = {cs, cm, cl};
  const uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t base = blockIdx.x * blockDim.x; base < owned;
       base += stride) {
    if (threadIdx.x < 3) s_cnt[threadIdx.x] = 0;
    __syncthreads();
    uint32_t r = base + threadIdx.x;
    int b = -1;
    uint32_t loc = 0;
    if (r < owned) {
      uint64_t deg = off[r + 1] - off[r];
      b = deg < kSmallDeg ? 0 : (deg < kMidDeg ? 1 : 2);
      loc = atomicAdd(&s_cnt[b], 1u);
    }
    __syncthreads();
    if (threadIdx.x < 3 && s_cnt[threadIdx.x])
      s_base[threadIdx.x] =
          atomicAdd(gcnt[threadIdx.x],
                    static_cast<unsigned long long>(s_cnt[threadIdx.x]));
    __syncthreads();
    if (b >= 0) lists[b][s_base[b] + loc] = r;
    __syncthreads();
  }
}

// Phase-split pulls (multi-GPU compute/comm overlap): the per-row
// adjacency is sorted ascending, so the owned dst range [lo,hi) is one
// contiguous span found by two binary searches. PHASE 0 = whole row
// (single-GPU / no overlap); PHASE 1 = owned span only, overwrite acc
// (runs WHILE the mirror exchange moves remote contribs on the comm
// stream); PHASE 2 = remaining spans, accumulate into acc.
__device__ __forceinline__ uint64_t lb_u32(const uint32_t* __restrict__ a,
                                           uint64_t lo, uint64_t hi,
                                           uint32_t key) {
  while (lo < hi) {
    uint64_t m = (lo + hi) >> 1;
    if (a[m] < key) lo = m + 1;
    else hi = m;
  }
  return lo;
}

// maps a phase to an iteration domain over row [b,e): phase 1 iterates
// [L,R); phase 2 iterates [b,L) ++ [R,e) via a folded index.
struct PrSpan {
  uint64_t base1, n1, base2, n2;
  __device__ __forceinline__ uint64_t map(uint64_t k) const {
    return k < n1 ? base1 + k : base2 + (k - n1);
  }
  __device__ __forceinline__ uint64_t n() const { return n1 + n2; }
};

template <int PHASE>
__device__ __forceinline__ PrSpan pr_span(const uint32_t* __restrict__ dst,
                                          uint64_t b, uint64_t e,
                                          uint32_t lo, uint32_t hi) {
  if (PHASE == 0) return {b, e - b, 0, 0};
  uint64_t L = lb_u32(dst, b, e, lo), R = lb_u32(dst, b, e, hi);
  if (PHASE == 1) return {L, R - L, 0, 0};
  return {b, L - b, R, e - R};
}

// Edge-parallel small-row pull (CM idiom): thread-per-row streaming left
// each lane walking its own row — scattered 64B lines per 4B edge read.
// A block stages 256 short rows' spans in LDS, prefix-sums their
// lengths, then sweeps the chunk's edges with CONSECUTIVE lanes on
// consecutive edge slots (coalesced dst/contrib streams), wave-segmented
// sums per row and LDS fp64 accumulation at run tails.
template <int PHASE>
__global__ void pr_pull_small_cm_kernel(
    const uint64_t* __restrict__ off, const uint32_t* __restrict__ dst,
    const float* __restrict__ contrib, const uint32_t* __restrict__ rows,
    uint64_t nrows, uint32_t v_begin, uint32_t lo, uint32_t hi,
    double* __restrict__ acc) {
  __shared__ uint64_t s_b1[kBlock], s_b2[kBlock];
  __shared__ uint32_t s_n1[kBlock], s_pref[kBlock + 1];
  __shared__ double s_acc[kBlock];
  const int lane = threadIdx.x & 63;
  for (uint64_t chunk = static_cast<uint64_t>(blockIdx.x) * kBlock;
       chunk < nrows;
       chunk += static_cast<uint64_t>(gridDim.x) * kBlock) {
    uint64_t i = chunk + threadIdx.x;
    uint32_t n = 0;
    if (i < nrows) {
      uint32_t r = rows[i];
      PrSpan sp = pr_span<PHASE>(dst, off[r], off[r + 1], lo, hi);
      s_b1[threadIdx.x] = sp.base1;
      s_n1[threadIdx.x] = static_cast<uint32_t>(sp.n1);
      s_b2[threadIdx.x] = sp.base2;
      n = static_cast<uint32_t>(sp.n());
    }
    s_acc[threadIdx.x] = 0.0;
    s_pref[threadIdx.x] = n;
    __syncthreads();
    // block-exclusive scan of the 256 lengths (Hillis-Steele in LDS)
    for (int d = 1; d < kBlock; d <<= 1) {
      uint32_t v = s_pref[threadIdx.x];
      uint32_t up = threadIdx.x >= d ? s_pref[threadIdx.x - d] : 0;
      __syncthreads();
      s_pref[threadIdx.x] = v + up;
      __syncthreads();
    }
    uint32_t total = s_pref[kBlock - 1];
    // shift to exclusive
    uint32_t excl = threadIdx.x ? s_pref[threadIdx.x - 1] : 0;
    __syncthreads();
    s_pref[threadIdx.x] = excl;
    if (threadIdx.x == 0) s_pref[kBlock] = total;
    __syncthreads();
    for (uint32_t base = 0; base < total; base += kBlock) {
      uint32_t e = base + threadIdx.x;
      bool act = e < total;
      uint32_t owner = 0xFFFFFFFFu;
      double c = 0.0;
      if (act) {
        uint32_t a = 0, b = kBlock;  // last owner with pref <= e
        while (a + 1 < b) {
          uint32_t m = (a + b) >> 1;
          if (s_pref[m] <= e) a = m;
          else b = m;
        }
        owner = a;
        uint32_t k = e - s_pref[a];
        uint64_t idx = k < s_n1[a]
                           ? s_b1[a] + k
                           : s_b2[a] + (k - s_n1[a]);
        c = static_cast<double>(contrib[dst[idx]]);
      }
      // wave-segmented sum keyed by owner (runs are contiguous)
      uint32_t oprev = __shfl_up(owner, 1, 64);
      bool head = (lane == 0) || oprev != owner;
      unsigned long long hb = __ballot(head);
      unsigned long long mine =
          hb & ((lane == 63) ? ~0ull : ((1ull << (lane + 1)) - 1));
      int seg = 63 - __clzll(mine);
      double sum = c;
#pragma unroll
      for (int d = 1; d < 64; d <<= 1) {
        double up = __shfl_up(sum, d, 64);
        if (lane - d >= seg) sum += up;
      }
      bool tail = (lane == 63) || ((hb >> (lane + 1)) & 1ull);
      if (tail && act) atomicAdd(&s_acc[owner], sum);
    }
    __syncthreads();
    if (i < nrows) {
      uint32_t r = rows[i];
      if (PHASE == 2) acc[v_begin + r] += s_acc[threadIdx.x];
      else acc[v_begin + r] = s_acc[threadIdx.x];
    }
    __syncthreads();
  }
}

// thread per row
template <int PHASE>
__global__ void pr_pull_small_kernel(const uint64_t* __restrict__ off,
                                     const uint32_t* __restrict__ dst,
                                     const float* __restrict__ contrib,
                                     const uint32_t* __restrict__ rows,
                                     uint64_t nrows, uint32_t v_begin,
                                     uint32_t lo, uint32_t hi,
                                     double* __restrict__ acc) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < nrows; i += stride) {
    uint32_t r = rows[i];
    PrSpan sp = pr_span<PHASE>(dst, off[r], off[r + 1], lo, hi);
    double sum = 0;
    for (uint64_t k = 0; k < sp.n(); ++k) sum += contrib[dst[sp.map(k)]];
    if (PHASE == 2) acc[v_begin + r] += sum;
    else acc[v_begin + r] = sum;
  }
}

// wave per row (4 waves per 256-block)
template <int PHASE>
__global__ void pr_pull_mid_kernel(const uint64_t* __restrict__ off,
                                   const uint32_t* __restrict__ dst,
                                   const float* __restrict__ contrib,
                                   const uint32_t* __restrict__ rows,
                                   uint64_t nrows, uint32_t v_begin,
                                   uint32_t lo, uint32_t hi,
                                   double* __restrict__ acc) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int waves_per_block = kBlock / kWave;
  size_t wstride = static_cast<size_t>(gridDim.x) * waves_per_block;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * waves_per_block + wid;
       i < nrows; i += wstride) {
    uint32_t r = rows[i];
    PrSpan sp = pr_span<PHASE>(dst, off[r], off[r + 1], lo, hi);
    double sum = 0;
    for (uint64_t k = lane; k < sp.n(); k += kWave)
      sum += contrib[dst[sp.map(k)]];
#pragma unroll
    for (int d = 32; d > 0; d >>= 1) sum += __shfl_down(sum, d, 64);
    if (lane == 0) {
      if (PHASE == 2) acc[v_begin + r] += sum;
      else acc[v_begin + r] = sum;
    }
  }
}

// block per row
template <int PHASE>
__global__ void pr_pull_large_kernel(const uint64_t* __restrict__ off,
                                     const uint32_t* __restrict__ dst,
                                     const float* __restrict__ contrib,
                                     const uint32_t* __restrict__ rows,
                                     uint64_t nrows, uint32_t v_begin,
                                     uint32_t lo, uint32_t hi,
                                     double* __restrict__ acc) {
  __shared__ double s_wave[kBlock / kWave];
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t r = rows[i];
    PrSpan sp = pr_span<PHASE>(dst, off[r], off[r + 1], lo, hi);
    double sum = 0;
    for (uint64_t k = threadIdx.x; k < sp.n(); k += blockDim.x)
      sum += contrib[dst[sp.map(k)]];
#pragma unroll
    for (int d = 32; d > 0; d >>= 1) sum += __shfl_down(sum, d, 64);
    if ((threadIdx.x & 63) == 0) s_wave[threadIdx.x >> 6] = sum;
    __syncthreads();
    if (threadIdx.x == 0) {
      double t = 0;
#pragma unroll
      for (int w = 0; w < kBlock / kWave; ++w) t += s_wave[w];
      if (PHASE == 2) acc[v_begin + r] += t;
      else acc[v_begin + r] = t;
    }
    __syncthreads();
  }
}

// ===========================================================================
// PageRank tiled pull. The row-major pull gathers contrib[dst] across the
// whole vertex range: at datagen-9_0 scale every 4-byte gather fetches a
// distinct 64-byte line (~6% line use, r01 PMC: 2.3 TB/s fetch, >80%
// latency stall). Regrouping edges by DST TILE (4096 vertices = 16 KB of
// contribs) makes consecutive gathers hit the same L2-resident window, so
// per-iteration traffic drops to ~(8 B/edge edge stream + one pass over
// contrib + L2-resident acc atomics). Rows' dst lists are sorted, so each
// row contributes contiguous per-tile runs; runs are placed contiguously
// inside their tile bucket (order across rows is irrelevant — each run
// has constant src, and a wave-segmented sum emits one fp64 atomic per
// run per 64-edge window). Built once per graph, cached in HBM3E
// (8 B/stored edge).
// ===========================================================================

constexpr int kPrTileBits = 12;  // 4096 vertices/tile: 16 KB fp32 window

// count (out==nullptr) or place (out!=nullptr) per-row tile runs.
// thread-per-row variant for short rows.
__global__ void pr_tile_runs_small_kernel(
    const uint64_t* __restrict__ off, const uint32_t* __restrict__ dst,
    const uint32_t* __restrict__ rows, uint64_t nrows, uint32_t v_begin,
    unsigned long long* __restrict__ cnt, unsigned long long* __restrict__ out) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < nrows; i += stride) {
    uint32_t r = rows[i];
    uint64_t b = off[r], e = off[r + 1];
    uint64_t src64 = static_cast<uint64_t>(v_begin + r) << 32;
    uint64_t k = b;
    while (k < e) {
      uint32_t t = dst[k] >> kPrTileBits;
      uint64_t j = k + 1;
      while (j < e && (dst[j] >> kPrTileBits) == t) ++j;
      unsigned long long pos = atomicAdd(&cnt[t], j - k);
      if (out)
        for (uint64_t q = k; q < j; ++q)
          out[pos + (q - k)] = src64 | dst[q];
      k = j;
    }
  }
}

// wave-per-row variant: lanes stride the row; run boundaries via ballot.
__global__ void pr_tile_runs_wave_kernel(
    const uint64_t* __restrict__ off, const uint32_t* __restrict__ dst,
    const uint32_t* __restrict__ rows, uint64_t nrows, uint32_t v_begin,
    unsigned long long* __restrict__ cnt, unsigned long long* __restrict__ out) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wpb = kBlock / kWave;
  size_t wstride = static_cast<size_t>(gridDim.x) * wpb;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * wpb + wid; i < nrows;
       i += wstride) {
    uint32_t r = rows[i];
    uint64_t b = off[r], e = off[r + 1];
    uint64_t src64 = static_cast<uint64_t>(v_begin + r) << 32;
    for (uint64_t base = b; base < e; base += kWave) {
      uint64_t k = base + lane;
      bool act = k < e;
      uint32_t d = act ? dst[k] : 0xFFFFFFFFu;
      uint32_t t = d >> kPrTileBits;
      // head: first lane of window, or tile differs from previous lane's
      uint32_t tprev = __shfl_up(t, 1, 64);
      bool head = act && (lane == 0 || tprev != t ||
                          (base + lane - 1) < b);
      // window runs may continue the previous window's run: lane0 head
      // still claims fresh space — runs split at window boundaries,
      // which only adds an extra atomic, not an error
      unsigned long long hb = __ballot(head);
      unsigned long long pos = 0;
      if (head && act) {
        // run length = next head (or window end / row end) - k
        unsigned long long later = hb >> 1 >> lane;  // heads after me
        int next = later ? lane + 1 + __ffsll(later) - 1 : 64;
        uint64_t rend = base + next;
        if (rend > e) rend = e;
        pos = atomicAdd(&cnt[t], rend - k);
      }
      if (out && act) {
        // my position: broadcast pos from my run's head lane
        unsigned long long mine =
            hb & ((lane == 63) ? ~0ull : ((1ull << (lane + 1)) - 1));
        int myhead = 63 - __clzll(mine);
        unsigned long long hpos =
            __shfl(pos, myhead, 64);
        out[hpos + (lane - myhead)] = src64 | d;
      }
    }
  }
}

// one pass over the tiled edge stream: gather contrib from the L2-hot
// window, wave-segmented sum per constant-src run, fp64 atomic per run.
__global__ void pr_tiled_pull_kernel(
    const unsigned long long* __restrict__ tiles, uint64_t lo, uint64_t hi,
    const float* __restrict__ contrib, double* __restrict__ acc) {
  const int lane = threadIdx.x & 63;
  uint64_t n = hi - lo;
  if (!n) return;
  // contiguous chunk per wave (keeps each wave inside one tile window)
  uint64_t waves = (static_cast<uint64_t>(gridDim.x) * blockDim.x) / kWave;
  uint64_t wid = (static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                  threadIdx.x) / kWave;
  uint64_t chunk = (n + waves - 1) / waves;
  uint64_t wb = lo + wid * chunk;
  uint64_t we = wb + chunk < hi ? wb + chunk : hi;
  for (uint64_t base = wb; base < we; base += kWave) {
    uint64_t k = base + lane;
    bool act = k < we;
    unsigned long long rec = act ? tiles[k] : ~0ull;
    uint32_t srcv = static_cast<uint32_t>(rec >> 32);
    double c = act ? static_cast<double>(contrib[static_cast<uint32_t>(rec)])
                   : 0.0;
    uint32_t sprev = __shfl_up(srcv, 1, 64);
    bool head = (lane == 0) || sprev != srcv;
    unsigned long long hb = __ballot(head);
    // segment start = highest head lane <= me
    unsigned long long mine =
        hb & ((lane == 63) ? ~0ull : ((1ull << (lane + 1)) - 1));
    int seg = 63 - __clzll(mine);
    double sum = c;
#pragma unroll
    for (int d = 1; d < 64; d <<= 1) {
      double up = __shfl_up(sum, d, 64);
      if (lane - d >= seg) sum += up;
    }
    bool tail = (lane == 63) || ((hb >> (lane + 1)) & 1ull);
    if (tail && act && srcv != 0xFFFFFFFFu) atomicAdd(&acc[srcv], sum);
  }
}


// ---------------------------------------------------------------------------
// Direction-optimizing BFS pull kernels (reference cuda/bfs/bfs.h :206-260):
// sweep unvisited owned rows looking for a parent at the current level.
// No atomics, early exit; reuses the degree buckets.
// ---------------------------------------------------------------------------
__global__ void bfs_pull_small_kernel(const uint64_t* __restrict__ off,
                                      const uint32_t* __restrict__ dst,
                                      const uint32_t* __restrict__ rows,
                                      uint64_t nrows, uint32_t v_begin,
                                      uint32_t* __restrict__ depth,
                                      uint32_t level, DevBitmap next_bm) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < nrows; i += stride) {
    uint32_t r = rows[i];
    uint32_t v = v_begin + r;
    if (depth[v] != 0xFFFFFFFFu) continue;
    uint64_t b = off[r], e = off[r + 1];
    for (uint64_t k = b; k < e; ++k) {
      if (depth[dst[k]] == level) {
        depth[v] = level + 1;
        next_bm.set_once(r);
        break;
      }
    }
  }
}

__global__ void bfs_pull_mid_kernel(const uint64_t* __restrict__ off,
                                    const uint32_t* __restrict__ dst,
                                    const uint32_t* __restrict__ rows,
                                    uint64_t nrows, uint32_t v_begin,
                                    uint32_t* __restrict__ depth,
                                    uint32_t level, DevBitmap next_bm) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wpb = kBlock / kWave;
  size_t wstride = static_cast<size_t>(gridDim.x) * wpb;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * wpb + wid; i < nrows;
       i += wstride) {
    uint32_t r = rows[i];
    uint32_t v = v_begin + r;
    if (depth[v] != 0xFFFFFFFFu) continue;
    uint64_t b = off[r], e = off[r + 1];
    bool found = false;
    for (uint64_t kb = b; kb < e; kb += kWave) {
      uint64_t k = kb + lane;
      bool mine = k < e && depth[dst[k]] == level;
      if (__any(mine)) {
        found = true;
        break;
      }
    }
    if (found && lane == 0) {
      depth[v] = level + 1;
      next_bm.set_once(r);
    }
  }
}

__global__ void bfs_pull_large_kernel(const uint64_t* __restrict__ off,
                                      const uint32_t* __restrict__ dst,
                                      const uint32_t* __restrict__ rows,
                                      uint64_t nrows, uint32_t v_begin,
                                      uint32_t* __restrict__ depth,
                                      uint32_t level, DevBitmap next_bm) {
  __shared__ int s_found;
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t r = rows[i];
    uint32_t v = v_begin + r;
    if (depth[v] != 0xFFFFFFFFu) continue;
    uint64_t b = off[r], e = off[r + 1];
    if (threadIdx.x == 0) s_found = 0;
    __syncthreads();
    for (uint64_t kb = b; kb < e; kb += blockDim.x) {
      uint64_t k = kb + threadIdx.x;
      if (k < e && depth[dst[k]] == level) s_found = 1;
      __syncthreads();
      if (s_found) break;
      __syncthreads();
    }
    if (threadIdx.x == 0 && s_found) {
      depth[v] = level + 1;
      next_bm.set_once(r);
    }
    __syncthreads();
  }
}

// ===========================================================================
// GpuContext implementation
// ===========================================================================

// ===========================================================================
// Device data plane. Production path: RCCL over xGMI, one rank per GPU
// (reference gpu_message_manager.h NCCL ring, RCCL is the MI355X drop-in).
// Validation path: TCP-staged collectives for when several ranks share one
// device (RCCL refuses duplicate GPUs) — selected by GRAPEHIP_DATAPLANE=tcp
// or auto-detected at init. Both paths drive the IDENTICAL call sites, so
// the world-N halo/allgather/reduce-scatter logic is exercised end-to-end
// on a one-GPU lease before an 8-GPU node ever sees it.
// ===========================================================================
struct DevComm {
  ncclComm_t nccl = nullptr;
  TcpComm* tcp = nullptr;
  int rank = 0, world = 1;
  bool use_tcp = false;
  // per-rank data-plane byte accounting (this rank's sends), split into
  // point-to-point (halo/mirror/row-fetch) and collective traffic —
  // read back per app run so comm volume vs boundary size is observable
  uint64_t bytes_p2p = 0, bytes_coll = 0;

  bool active() const { return world > 1; }

  // --- tcp staging helpers (host bounce; correctness path, not perf) ------
  std::string stage_out(const void* dev, size_t n, hipStream_t s) const {
    std::string h(n, '\0');
    HIP_CHECK(hipStreamSynchronize(s));
    if (n) HIP_CHECK(hipMemcpy(h.data(), dev, n, hipMemcpyDeviceToHost));
    return h;
  }
  void stage_in(void* dev, const void* host, size_t n) const {
    if (n) HIP_CHECK(hipMemcpy(dev, host, n, hipMemcpyHostToDevice));
  }

  // In-place allgather of 4-byte slices: buf[rank*slice .. ) is this rank's
  // contribution; on return buf[0 .. world*slice) holds every rank's.
  void allgather32(void* buf, uint64_t slice, bool fp, hipStream_t s) {
    if (!active()) return;
    bytes_coll += slice * 4;
    if (!use_tcp) {
      NCCL_CHECK(ncclAllGather(
          static_cast<char*>(buf) + static_cast<uint64_t>(rank) * slice * 4,
          buf, slice, fp ? ncclFloat : ncclUint32, nccl, s));
      return;
    }
    std::string mine = stage_out(
        static_cast<char*>(buf) + static_cast<uint64_t>(rank) * slice * 4,
        slice * 4, s);
    std::vector<char> all(static_cast<size_t>(world) * slice * 4);
    tcp->allgather(mine.data(), slice * 4, all.data());
    stage_in(buf, all.data(), all.size());
  }

  // In-place reduce-scatter (sum, f64): buf holds world*slice doubles; on
  // return buf[rank*slice ..) = elementwise sum of every rank's slice.
  void reduce_scatter_sum_f64(double* buf, uint64_t slice, hipStream_t s) {
    if (!active()) return;
    bytes_coll += static_cast<uint64_t>(world) * slice * 8;
    if (!use_tcp) {
      NCCL_CHECK(ncclReduceScatter(
          buf, buf + static_cast<uint64_t>(rank) * slice, slice, ncclDouble,
          ncclSum, nccl, s));
      return;
    }
    std::string full =
        stage_out(buf, static_cast<size_t>(world) * slice * 8, s);
    const double* h = reinterpret_cast<const double*>(full.data());
    std::vector<std::string> send(world);
    for (int p = 0; p < world; ++p)
      send[p].assign(
          reinterpret_cast<const char*>(h + static_cast<uint64_t>(p) * slice),
          slice * 8);
    auto recv = tcp->exchange_all(send);
    std::vector<double> acc(slice, 0.0);
    for (int p = 0; p < world; ++p) {
      const double* rp = reinterpret_cast<const double*>(recv[p].data());
      for (uint64_t i = 0; i < slice; ++i) acc[i] += rp[i];
    }
    stage_in(buf + static_cast<uint64_t>(rank) * slice, acc.data(),
             slice * 8);
  }

  void allreduce_min_u32(uint32_t* buf, uint64_t n, hipStream_t s) {
    if (!active()) return;
    bytes_coll += n * 4;
    if (!use_tcp) {
      NCCL_CHECK(
          ncclAllReduce(buf, buf, n, ncclUint32, ncclMin, nccl, s));
      return;
    }
    std::string mine = stage_out(buf, n * 4, s);
    std::vector<uint32_t> all(static_cast<size_t>(world) * n);
    tcp->allgather(mine.data(), n * 4, all.data());
    uint32_t* acc = reinterpret_cast<uint32_t*>(mine.data());
    for (int p = 0; p < world; ++p) {
      const uint32_t* rp = all.data() + static_cast<uint64_t>(p) * n;
      for (uint64_t i = 0; i < n; ++i)
        if (rp[i] < acc[i]) acc[i] = rp[i];
    }
    stage_in(buf, acc, n * 4);
  }

  void allreduce_sum_u64(unsigned long long* buf, uint64_t n,
                         hipStream_t s) {
    if (!active()) return;
    bytes_coll += n * 8;
    if (!use_tcp) {
      NCCL_CHECK(ncclAllReduce(buf, buf, n, ncclUint64, ncclSum, nccl, s));
      return;
    }
    std::string mine = stage_out(buf, n * 8, s);
    std::vector<unsigned long long> all(static_cast<size_t>(world) * n);
    tcp->allgather(mine.data(), n * 8, all.data());
    unsigned long long* acc =
        reinterpret_cast<unsigned long long*>(mine.data());
    for (uint64_t i = 0; i < n; ++i) acc[i] = 0;
    for (int p = 0; p < world; ++p) {
      const unsigned long long* rp =
          all.data() + static_cast<uint64_t>(p) * n;
      for (uint64_t i = 0; i < n; ++i) acc[i] += rp[i];
    }
    stage_in(buf, acc, n * 8);
  }

  // Pairwise variable-size exchange: send_off/recv_off are [world+1] byte
  // offsets into sendbuf/recvbuf; peer p receives this rank's region p.
  // Self region is handled by the caller (halo appends skip owned ids).
  void sendrecv(const uint8_t* sendbuf, const std::vector<uint64_t>& send_off,
                uint8_t* recvbuf, const std::vector<uint64_t>& recv_off,
                hipStream_t s) {
    if (!active()) return;
    for (int p = 0; p < world; ++p)
      if (p != rank) bytes_p2p += send_off[p + 1] - send_off[p];
    if (!use_tcp) {
      NCCL_CHECK(ncclGroupStart());
      for (int p = 0; p < world; ++p) {
        if (p == rank) continue;
        uint64_t to_p = send_off[p + 1] - send_off[p];
        uint64_t from_p = recv_off[p + 1] - recv_off[p];
        if (to_p)
          NCCL_CHECK(ncclSend(sendbuf + send_off[p], to_p, ncclChar, p,
                              nccl, s));
        if (from_p)
          NCCL_CHECK(ncclRecv(recvbuf + recv_off[p], from_p, ncclChar, p,
                              nccl, s));
      }
      NCCL_CHECK(ncclGroupEnd());
      return;
    }
    HIP_CHECK(hipStreamSynchronize(s));
    std::vector<std::string> send(world);
    for (int p = 0; p < world; ++p) {
      if (p == rank) continue;
      uint64_t n = send_off[p + 1] - send_off[p];
      send[p] = stage_out(sendbuf + send_off[p], n, s);
    }
    auto recv = tcp->exchange_all(send);
    for (int p = 0; p < world; ++p) {
      if (p == rank) continue;
      uint64_t n = recv_off[p + 1] - recv_off[p];
      if (n != recv[p].size())
        throw std::runtime_error("sendrecv(tcp): size mismatch");
      stage_in(recvbuf + recv_off[p], recv[p].data(), n);
    }
  }

  // Ragged in-place allgather: rank f owns buf[region[f]*4 .. region[f+1]*4)
  // (u32 elements); every rank ends with all regions populated.
  void bcast_regions_u32(uint32_t* buf, const std::vector<uint64_t>& region,
                         hipStream_t s) {
    if (!active()) return;
    bytes_coll += (region[rank + 1] - region[rank]) * 4;
    if (!use_tcp) {
      NCCL_CHECK(ncclGroupStart());
      for (int f = 0; f < world; ++f) {
        uint64_t cnt_f = region[f + 1] - region[f];
        if (cnt_f)
          NCCL_CHECK(ncclBroadcast(buf + region[f], buf + region[f], cnt_f,
                                   ncclUint32, f, nccl, s));
      }
      NCCL_CHECK(ncclGroupEnd());
      return;
    }
    uint64_t mine_n = region[rank + 1] - region[rank];
    std::string mine = stage_out(buf + region[rank], mine_n * 4, s);
    std::vector<std::string> send(world, mine);
    auto recv = tcp->exchange_all(send);
    for (int f = 0; f < world; ++f) {
      if (f == rank) continue;
      uint64_t n = region[f + 1] - region[f];
      if (n * 4 != recv[f].size())
        throw std::runtime_error("bcast_regions(tcp): size mismatch");
      stage_in(buf + region[f], recv[f].data(), n * 4);
    }
  }
};

struct GpuContext::Impl {
  DevComm dc;
  Stream compute;
  Stream comm_stream;
  Event ev_pack;   // compute -> comm ordering for the halo exchange
  Event ev_comm;   // comm -> compute ordering (recv payload ready)
  Event ev_scal;   // pinned-scalar D2H landed (host folds while GPU works)
  double* h_scal = nullptr;  // 2 pinned doubles: [0] local out, [1] global in
  Impl() { HIP_CHECK(hipHostMalloc(&h_scal, 2 * sizeof(double))); }
  ~Impl() {
    if (h_scal) (void)hipHostFree(h_scal);
  }
  ScanTemp scan;
  // halo scratch (sized on first use)
  DeviceBuffer<uint32_t> halo_idx;
  DeviceBuffer<unsigned long long> halo_cnt;
  DeviceBuffer<uint32_t> halo_bm;
  DeviceBuffer<uint8_t> sendbuf, recvbuf;
  // frontier scratch
  DeviceBuffer<uint32_t> frontier_deg;
  DeviceBuffer<uint64_t> frontier_off;
  double t_exchange = 0;
};

namespace {

DevGraphView make_view(const DeviceGraph& g, int rank, int world) {
  DevGraphView v{};
  v.nv_global = g.nv_global;
  v.v_begin = g.v_begin;
  v.v_end = g.v_end;
  v.slice = g.seg_host.size() > 1 ? (g.seg_host[1] - g.seg_host[0])
                                  : g.nv_global;
  if (v.slice == 0) v.slice = 1;
  v.rank = rank;
  v.world = world;
  v.oe_off = g.oe_off.data();
  v.oe_dst = g.oe_dst.data();
  v.oe_w = g.weighted ? g.oe_w.data() : nullptr;
  v.ie_off = g.has_in ? g.ie_off.data() : nullptr;
  v.ie_dst = g.has_in ? g.ie_dst.data() : nullptr;
  v.ie_w = (g.has_in && g.weighted) ? g.ie_w.data() : nullptr;
  v.seg = g.seg.data();
  return v;
}

uint32_t padded_nv(const DeviceGraph& g, int world) {
  uint32_t slice = g.seg_host.size() > 1 ? (g.seg_host[1] - g.seg_host[0])
                                         : g.nv_global;
  uint64_t pad = static_cast<uint64_t>(slice) * world;
  return static_cast<uint32_t>(pad > g.nv_global ? pad : g.nv_global);
}

double wall_s() {
  return std::chrono::duration<double>(
             std::chrono::steady_clock::now().time_since_epoch())
      .count();
}

}  // namespace

DeviceGraph::~DeviceGraph() {
  if (pr_graph_exec)
    (void)hipGraphExecDestroy(
        reinterpret_cast<hipGraphExec_t>(pr_graph_exec));
}

GpuContext::GpuContext(TcpComm* comm, int rank, int world)
    : comm_(comm), rank_(rank), world_(world) {
  int ndev = 0;
  HIP_CHECK(hipGetDeviceCount(&ndev));
  if (ndev == 0) throw std::runtime_error("no HIP devices visible");
  const char* lr = std::getenv("LOCAL_RANK");
  dev_ = lr ? (std::atoi(lr) % ndev) : (rank % ndev);
  HIP_CHECK(hipSetDevice(dev_));
  impl_ = std::make_unique<Impl>();
  DevComm& dc = impl_->dc;
  dc.tcp = comm;
  dc.rank = rank;
  dc.world = world;
  if (world > 1) {
    // data-plane pick: RCCL needs one distinct GPU per rank; ranks that
    // share a device (world-N validation on a 1-GPU lease) stage through
    // the TCP control plane instead. Same call sites either way.
    const char* dp = std::getenv("GRAPEHIP_DATAPLANE");
    if (dp && std::string(dp) == "tcp") {
      dc.use_tcp = true;
    } else if (dp && std::string(dp) == "rccl") {
      dc.use_tcp = false;
    } else {
      std::vector<int> devs(world);
      comm->allgather(&dev_, sizeof(int), devs.data());
      std::sort(devs.begin(), devs.end());
      dc.use_tcp = std::adjacent_find(devs.begin(), devs.end()) !=
                   devs.end();
    }
    if (!dc.use_tcp) {
      ncclUniqueId id;
      if (rank == 0) NCCL_CHECK(ncclGetUniqueId(&id));
      comm->bcast(&id, sizeof(id), 0);
      NCCL_CHECK(ncclCommInitRank(&dc.nccl, world, id, rank));
      // warm up the xGMI links (reference WarmupNccl, dev_utils.h:894)
      DeviceBuffer<float> dummy(world * 256);
      NCCL_CHECK(ncclAllReduce(dummy.data(), dummy.data(), 256, ncclFloat,
                               ncclSum, dc.nccl, impl_->compute));
      impl_->compute.sync();
    } else if (getenv("GRAPEHIP_DEBUG")) {
      fprintf(stderr, "[dc] rank %d: TCP data plane (shared device)\n",
              rank);
    }
  }
}

GpuContext::~GpuContext() {
  if (impl_ && impl_->dc.nccl) ncclCommDestroy(impl_->dc.nccl);
}

void GpuContext::device_sync() { HIP_CHECK(hipDeviceSynchronize()); }

std::vector<uint64_t> GpuContext::debug_scan(const std::vector<uint32_t>& in) {
  hipStream_t s = impl_->compute;
  DeviceBuffer<uint32_t> d_in;
  d_in.upload(in, s);
  DeviceBuffer<uint64_t> d_out(in.size() + 1);
  exclusive_scan(d_in.data(), d_out.data(), in.size(), s, impl_->scan);
  return d_out.download(s);
}

// ---------------------------------------------------------------------------
// Graph construction
// ---------------------------------------------------------------------------

namespace {

void build_csr_from_coo(const DeviceBuffer<uint32_t>& src,
                        const DeviceBuffer<uint32_t>& dst,
                        const DeviceBuffer<float>& w, uint64_t n,
                        uint32_t v_begin, uint32_t owned, bool weighted,
                        DeviceBuffer<uint64_t>& out_off,
                        DeviceBuffer<uint32_t>& out_dst,
                        DeviceBuffer<float>& out_w, hipStream_t s,
                        ScanTemp& scan) {
  DeviceBuffer<uint32_t> deg(owned);
  deg.zero(s);
  if (n)
    count_deg_kernel<<<grid_for(n), kBlock, 0, s>>>(src.data(), n, v_begin,
                                                    deg.data());
  out_off.resize(owned + 1);
  uint64_t total = exclusive_scan(deg.data(), out_off.data(), owned, s, scan);
  if (getenv("GRAPEHIP_DEBUG"))
    fprintf(stderr, "[csr] scan total=%lu n=%lu\n", (unsigned long)total,
            (unsigned long)n);
  if (total != n) throw std::runtime_error("CSR build: degree sum mismatch");
  deg.free();
  DeviceBuffer<unsigned long long> cursor(owned);
  HIP_CHECK(hipMemcpyAsync(cursor.data(), out_off.data(), owned * 8,
                           hipMemcpyDeviceToDevice, s));
  out_dst.resize(total);
  if (weighted) out_w.resize(total);
  if (n)
    scatter_edges_kernel<<<grid_for(n), kBlock, 0, s>>>(
        src.data(), dst.data(), weighted ? w.data() : nullptr, n, v_begin,
        cursor.data(), out_dst.data(), weighted ? out_w.data() : nullptr);
  HIP_CHECK(hipStreamSynchronize(s));
}

void sort_csr_rows(const DeviceBuffer<uint64_t>& off,
                   DeviceBuffer<uint32_t>& dst, DeviceBuffer<float>* w,
                   uint32_t rows_n, hipStream_t s, ScanTemp& scan) {
  if (rows_n == 0) return;
  DeviceBuffer<uint32_t> lds_rows(rows_n), big_rows(rows_n);
  DeviceBuffer<unsigned long long> cnts(2);
  cnts.zero(s);
  seg_sortbucket_kernel<<<grid_for(rows_n), kBlock, 0, s>>>(
      off.data(), rows_n, lds_rows.data(), cnts.data() + 0, big_rows.data(),
      cnts.data() + 1);
  auto h = cnts.download(s);
  if (h[0])
    seg_sort_lds_kernel<<<std::min<uint64_t>(h[0], kMaxGrid), kBlock, 0,
                          s>>>(off.data(), dst.data(),
                               w ? w->data() : nullptr, lds_rows.data(),
                               h[0]);
  if (h[1]) {
    DeviceBuffer<uint32_t> caps(h[1]);
    DeviceBuffer<uint64_t> pad_off(h[1] + 1);
    seg_bigcap_kernel<<<grid_for(h[1]), kBlock, 0, s>>>(
        off.data(), big_rows.data(), h[1], caps.data());
    uint64_t pad_total =
        exclusive_scan(caps.data(), pad_off.data(), h[1], s, scan);
    DeviceBuffer<unsigned long long> scratch(pad_total);
    seg_bigpad_kernel<<<std::min<uint64_t>(h[1], kMaxGrid), kBlock, 0, s>>>(
        off.data(), dst.data(), w ? w->data() : nullptr, big_rows.data(),
        h[1], pad_off.data(), scratch.data());
    seg_bigsort_kernel<<<std::min<uint64_t>(h[1], kMaxGrid), kBlock, 0,
                         s>>>(pad_off.data(), h[1], scratch.data());
    seg_bigunpad_kernel<<<std::min<uint64_t>(h[1], kMaxGrid), kBlock, 0,
                          s>>>(off.data(), dst.data(),
                               w ? w->data() : nullptr, big_rows.data(),
                               h[1], pad_off.data(), scratch.data());
  }
  HIP_CHECK(hipStreamSynchronize(s));
}

}  // namespace

std::unique_ptr<DeviceGraph> GpuContext::gen_synthetic(
    uint64_t nv, uint64_t ne, uint64_t seed, bool directed, bool weighted,
    bool build_in_csr, double a, double b, double c) {
  if (nv >= (1ull << 32))
    throw std::runtime_error("gen_synthetic: nv must fit in 32 bits");
  auto g = std::make_unique<DeviceGraph>();
  g->nv_global = static_cast<uint32_t>(nv);
  g->nv_real = nv;
  g->directed = directed;
  g->weighted = weighted;
  g->input_edges = ne;
  uint32_t slice = static_cast<uint32_t>((nv + world_ - 1) / world_);
  g->seg_host.resize(world_ + 1);
  for (int f = 0; f <= world_; ++f)
    g->seg_host[f] = static_cast<uint32_t>(
        std::min<uint64_t>(static_cast<uint64_t>(f) * slice, nv));
  g->seg.upload(g->seg_host, impl_->compute);
  g->v_begin = g->seg_host[rank_];
  g->v_end = g->seg_host[rank_ + 1];
  uint32_t owned = g->owned();
  g->owned_real = owned;

  int scale = 0;
  while ((1ull << scale) < nv) ++scale;
  uint32_t t_a = static_cast<uint32_t>(a * 65536.0);
  uint32_t t_ab = static_cast<uint32_t>((a + b) * 65536.0);
  uint32_t t_abc = static_cast<uint32_t>((a + b + c) * 65536.0);
  // slice-balance scramble (see gen_edges_kernel); deterministic per nv
  uint64_t scr_a = 0, scr_b = 0;
  if (!(getenv("GRAPEHIP_SCRAMBLE") &&
        atoi(getenv("GRAPEHIP_SCRAMBLE")) == 0) && nv > 2) {
    scr_a = static_cast<uint64_t>(nv * 0.6180339887498949);
    if (scr_a < 2) scr_a = 2;
    auto gcd_u64 = [](uint64_t x, uint64_t y) {
      while (y) {
        uint64_t t = x % y;
        x = y;
        y = t;
      }
      return x;
    };
    while (gcd_u64(scr_a, nv) != 1) ++scr_a;
    scr_b = (seed * 0x9E3779B97F4A7C15ULL) % nv;
  }

  hipStream_t s = impl_->compute;
  // exact per-rank edge count via a counting pass (see gen_edges_kernel)
  DeviceBuffer<unsigned long long> cnt(1);
  cnt.zero(s);
  uint64_t est;
  if (world_ == 1) {
    est = static_cast<uint64_t>((directed ? 1.0 : 2.0) * ne) + 16;
  } else {
    gen_edges_kernel<<<kMaxGrid, kBlock, 0, s>>>(
        ne, seed, scale, g->nv_global, t_a, t_ab, t_abc, scr_a, scr_b,
        g->v_begin, g->v_end, !directed, false, false, nullptr, nullptr,
        nullptr, cnt.data());
    unsigned long long c = 0;
    HIP_CHECK(hipMemcpyAsync(&c, cnt.data(), 8, hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    est = c + 16;
    cnt.zero(s);
  }
  DeviceBuffer<uint32_t> e_src(est), e_dst(est);
  DeviceBuffer<float> e_w(weighted ? est : 0);
  if (getenv("GRAPEHIP_DEBUG"))
    fprintf(stderr, "[gen] nv=%u ne=%lu est=%lu scale=%d\n", g->nv_global,
            (unsigned long)ne, (unsigned long)est, scale);
  gen_edges_kernel<<<kMaxGrid, kBlock, 0, s>>>(
      ne, seed, scale, g->nv_global, t_a, t_ab, t_abc, scr_a, scr_b, g->v_begin, g->v_end,
      !directed, false, weighted, e_src.data(), e_dst.data(),
      weighted ? e_w.data() : nullptr, cnt.data());
  unsigned long long n_local = 0;
  HIP_CHECK(hipMemcpyAsync(&n_local, cnt.data(), 8, hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipStreamSynchronize(s));
  if (n_local > est) throw std::runtime_error("gen_synthetic: overflow");
  if (getenv("GRAPEHIP_DEBUG"))
    fprintf(stderr, "[gen] n_local=%llu\n", n_local);

  // hub-clustering renumber: order each slice by descending appearance
  // count (out-stream src count = how often an id is REFERENCED by pull
  // gathers: full degree when undirected, out-degree when directed).
  // Counting sort over capped degrees; within-bucket order is arbitrary.
  const bool renumber =
      !(getenv("GRAPEHIP_RENUMBER") &&
        atoi(getenv("GRAPEHIP_RENUMBER")) == 0);
  if (renumber) {
    uint32_t nv_pad = padded_nv(*g, world_);
    DeviceBuffer<uint32_t> deg(owned ? owned : 1);
    deg.zero(s);
    if (n_local)
      count_deg_kernel<<<grid_for(n_local), kBlock, 0, s>>>(
          e_src.data(), n_local, g->v_begin, deg.data());
    DeviceBuffer<unsigned long long> hist(kRenumBuckets);
    hist.zero(s);
    if (owned)
      renum_hist_kernel<<<grid_for(owned), kBlock, 0, s>>>(
          deg.data(), owned, hist.data());
    auto h = hist.download(s);
    // descending bucket offsets (bucket 4095 = hottest, first)
    std::vector<unsigned long long> cur_h(kRenumBuckets);
    unsigned long long run = 0;
    for (int b = kRenumBuckets - 1; b >= 0; --b) {
      cur_h[b] = run;
      run += h[b];
    }
    DeviceBuffer<unsigned long long> cur;
    cur.upload(cur_h, s);
    g->perm.resize(nv_pad);
    iota_kernel<<<grid_for(nv_pad), kBlock, 0, s>>>(g->perm.data(), 0,
                                                    nv_pad);
    if (owned)
      renum_scatter_kernel<<<grid_for(owned), kBlock, 0, s>>>(
          deg.data(), owned, g->v_begin, cur.data(), g->perm.data());
    if (world_ > 1)
      impl_->dc.allgather32(g->perm.data(), slice, false, s);
    if (n_local) {
      remap_ids_kernel<<<grid_for(n_local), kBlock, 0, s>>>(
          e_src.data(), n_local, g->perm.data());
      remap_ids_kernel<<<grid_for(n_local), kBlock, 0, s>>>(
          e_dst.data(), n_local, g->perm.data());
    }
    g->inv.resize(nv_pad);
    inv_scatter_kernel<<<grid_for(nv_pad), kBlock, 0, s>>>(
        g->perm.data(), nv_pad, g->inv.data());
    g->permuted = true;
  }

  build_csr_from_coo(e_src, e_dst, e_w, n_local, g->v_begin, owned, weighted,
                     g->oe_off, g->oe_dst, g->oe_w, s, impl_->scan);
  sort_csr_rows(g->oe_off, g->oe_dst, weighted ? &g->oe_w : nullptr, owned,
                s, impl_->scan);
  if (directed && build_in_csr) {
    // regenerate the SAME edge stream, keeping edges whose dst is owned,
    // reversed — builds the in-CSR without materializing the global list
    if (world_ > 1) {
      // in-edge ownership skews differently than out; recount exactly
      cnt.zero(s);
      gen_edges_kernel<<<kMaxGrid, kBlock, 0, s>>>(
          ne, seed, scale, g->nv_global, t_a, t_ab, t_abc, scr_a, scr_b, g->v_begin,
          g->v_end, false, true, false, nullptr, nullptr, nullptr,
          cnt.data());
      unsigned long long c_in = 0;
      HIP_CHECK(hipMemcpyAsync(&c_in, cnt.data(), 8, hipMemcpyDeviceToHost,
                               s));
      HIP_CHECK(hipStreamSynchronize(s));
      if (c_in + 16 > est) {
        est = c_in + 16;
        e_src.free();
        e_dst.free();
        e_w.free();
        e_src.resize(est);
        e_dst.resize(est);
        if (weighted) e_w.resize(est);
      }
    }
    cnt.zero(s);
    gen_edges_kernel<<<kMaxGrid, kBlock, 0, s>>>(
        ne, seed, scale, g->nv_global, t_a, t_ab, t_abc, scr_a, scr_b, g->v_begin,
        g->v_end, false, true, weighted, e_src.data(), e_dst.data(),
        weighted ? e_w.data() : nullptr, cnt.data());
    unsigned long long n_in = 0;
    HIP_CHECK(hipMemcpyAsync(&n_in, cnt.data(), 8, hipMemcpyDeviceToHost,
                             s));
    HIP_CHECK(hipStreamSynchronize(s));
    if (n_in > est) throw std::runtime_error("gen_synthetic: in overflow");
    if (g->permuted && n_in) {
      remap_ids_kernel<<<grid_for(n_in), kBlock, 0, s>>>(
          e_src.data(), n_in, g->perm.data());
      remap_ids_kernel<<<grid_for(n_in), kBlock, 0, s>>>(
          e_dst.data(), n_in, g->perm.data());
    }
    build_csr_from_coo(e_src, e_dst, e_w, n_in, g->v_begin, owned, weighted,
                       g->ie_off, g->ie_dst, g->ie_w, s, impl_->scan);
    sort_csr_rows(g->ie_off, g->ie_dst, weighted ? &g->ie_w : nullptr,
                  owned, s, impl_->scan);
    g->has_in = true;
  }
  e_src.free();
  e_dst.free();
  e_w.free();
  g->local_edges = n_local;
  g->total_edges = comm_ && world_ > 1 ? comm_->allreduce_sum(n_local)
                                       : n_local;
  return g;
}

std::unique_ptr<DeviceGraph> GpuContext::upload(const Fragment& frag) {
  auto g = std::make_unique<DeviceGraph>();
  g->directed = frag.directed();
  g->weighted = frag.has_weights();
  g->has_in = frag.has_in_csr();
  g->input_edges = frag.input_edges();
  g->total_edges = frag.total_edges();
  g->local_edges = frag.local_edges();
  g->nv_real = frag.total_vertices();
  hipStream_t s = impl_->compute;
  const bool identity = frag.vm().idxer() == IdxerKind::kIdentity;

  // device id mapping. identity maps use oids directly (segmented
  // ownership from the vertex map). Arbitrary-oid (hashmap) maps are
  // DENSELY RENUMBERED on upload: device id = fid * slice + lid with one
  // uniform slice = max fragment vnum, so owner lookup stays v/slice and
  // RCCL slices stay uniform (the reference needs a device hashmap,
  // cuda/vertex_map/device_vertex_map.h, for the same capability).
  // Padding rows [ivnum, slice) get empty adjacency.
  std::function<uint32_t(vid_t)> lid2dev;
  uint32_t slice;
  if (identity) {
    const auto& seg64 = frag.vm().segments();
    g->seg_host.assign(seg64.begin(), seg64.end());
    g->nv_global = static_cast<uint32_t>(frag.total_vertices());
    g->v_begin = g->seg_host[frag.fid()];
    g->v_end = g->seg_host[frag.fid() + 1];
    slice = g->seg_host.size() > 1
                ? static_cast<uint32_t>(g->seg_host[1] - g->seg_host[0])
                : g->nv_global;
    lid2dev = [&frag](vid_t lid) {
      return static_cast<uint32_t>(frag.lid2oid(lid));
    };
  } else {
    uint64_t vmax = frag.ivnum();
    if (comm_ && world_ > 1) vmax = comm_->allreduce_max(vmax);
    slice = static_cast<uint32_t>(vmax);
    g->nv_global = static_cast<uint32_t>(
        static_cast<uint64_t>(slice) * world_);
    g->seg_host.resize(world_ + 1);
    for (int f = 0; f <= world_; ++f)
      g->seg_host[f] = static_cast<uint32_t>(
          static_cast<uint64_t>(f) * slice);
    g->v_begin = g->seg_host[frag.fid()];
    g->v_end = g->seg_host[frag.fid() + 1];
    const IdParser& P = frag.parser();
    lid2dev = [&frag, &P, slice](vid_t lid) {
      vid_t gid = frag.lid2gid(lid);
      return static_cast<uint32_t>(
          static_cast<uint64_t>(P.fid(gid)) * slice + P.lid(gid));
    };
  }
  g->owned_real = identity ? g->owned() : frag.ivnum();
  g->seg.upload(g->seg_host, s);

  auto convert = [&](const std::vector<vid_t>& dst_lid) {
    std::vector<uint32_t> out(dst_lid.size());
    parallel_for(0, dst_lid.size(), [&](size_t i) {
      out[i] = lid2dev(dst_lid[i]);
    }, 8192);
    return out;
  };
  auto pad_offsets = [&](const std::vector<eid_t>& off) {
    // extend to the padded owned range with empty rows
    uint32_t rows = g->v_end - g->v_begin;
    std::vector<eid_t> out(off);
    out.resize(rows + 1, off.empty() ? 0 : off.back());
    return out;
  };
  g->oe_off.upload(pad_offsets(frag.oe_offsets()), s);
  g->oe_dst.upload(convert(frag.oe_dsts()), s);
  if (g->weighted) g->oe_w.upload(frag.oe_weights(), s);
  if (g->has_in) {
    g->ie_off.upload(pad_offsets(frag.ie_offsets()), s);
    g->ie_dst.upload(convert(frag.ie_dsts()), s);
    if (g->weighted) g->ie_w.upload(frag.ie_weights(), s);
  }
  uint32_t owned_rows = g->owned();
  sort_csr_rows(g->oe_off, g->oe_dst, g->weighted ? &g->oe_w : nullptr,
                owned_rows, s, impl_->scan);
  if (g->has_in)
    sort_csr_rows(g->ie_off, g->ie_dst, g->weighted ? &g->ie_w : nullptr,
                  owned_rows, s, impl_->scan);
  if (!identity) {
    // CDLP tie-breaks compare label VALUES; the reference's labels are
    // oids, so under dense renumbering the comparison space must be the
    // global sorted-OID order (vertex map is replicated — every rank
    // derives the identical table)
    const VertexMap& vm = frag.vm();
    const IdParser& P = frag.parser();
    std::vector<std::pair<oid_t, uint32_t>> pairs;
    pairs.reserve(vm.total_vertices());
    for (int f = 0; f < vm.fnum(); ++f) {
      vid_t n = vm.frag_vnum(static_cast<fid_t>(f));
      for (vid_t l = 0; l < n; ++l)
        pairs.emplace_back(
            vm.get_oid(P.gid(static_cast<fid_t>(f), l)),
            static_cast<uint32_t>(static_cast<uint64_t>(f) * slice + l));
    }
    std::sort(pairs.begin(), pairs.end());
    std::vector<uint32_t> order(padded_nv(*g, world_), 0xFFFFFFFFu);
    for (uint32_t i = 0; i < pairs.size(); ++i)
      order[pairs[i].second] = i;
    g->oid_order.upload(order, s);
  }
  HIP_CHECK(hipStreamSynchronize(s));
  return g;
}

}  // namespace grapehip

namespace grapehip {

// ---------------------------------------------------------------------------
// Degree buckets over the pull CSR (undirected: out == neighborhood;
// directed: in-CSR). Built once per graph, reused by PR pull + BFS pull.
// ---------------------------------------------------------------------------
static void ensure_buckets(DeviceGraph& g, hipStream_t s) {
  if (g.buckets_built) return;
  uint32_t owned = g.owned();
  const uint64_t* pull_off = !g.directed ? g.oe_off.data() : g.ie_off.data();
  g.rows_small.resize(owned);
  g.rows_mid.resize(owned);
  g.rows_large.resize(owned);
  DeviceBuffer<unsigned long long> cnts(3);
  cnts.zero(s);
  bucket_rows_kernel<<<grid_for(owned), kBlock, 0, s>>>(
      pull_off, owned, g.rows_small.data(), cnts.data() + 0,
      g.rows_mid.data(), cnts.data() + 1, g.rows_large.data(),
      cnts.data() + 2);
  auto h = cnts.download(s);
  g.n_small = h[0];
  g.n_mid = h[1];
  g.n_large = h[2];
  g.buckets_built = true;
}

static void ensure_pr_tiles(DeviceGraph& g, GpuContext::Impl& I,
                            hipStream_t s) {
  if (g.pr_tiles_built) return;
  const uint64_t* pull_off = !g.directed ? g.oe_off.data() : g.ie_off.data();
  const uint32_t* pull_dst = !g.directed ? g.oe_dst.data() : g.ie_dst.data();
  uint64_t ne = !g.directed ? g.oe_dst.size() : g.ie_dst.size();
  uint32_t ntiles =
      (g.nv_global + (1u << kPrTileBits) - 1) >> kPrTileBits;
  // memory gate: 8 B/edge for the stream + counters; fall back to the
  // row-major pull when HBM is tight (huge 1B+-vertex graphs)
  size_t free_b = 0, total_b = 0;
  HIP_CHECK(hipMemGetInfo(&free_b, &total_b));
  if (free_b < ne * 8 + (static_cast<size_t>(ntiles) + 2) * 16 + (512u << 20)) {
    g.pr_tiles_built = true;  // decided: stay on fallback
    g.pr_ntiles = 0;
    return;
  }
  DeviceBuffer<unsigned long long> cnt(ntiles + 1);
  cnt.zero(s);
  auto runs = [&](unsigned long long* c, unsigned long long* out) {
    if (g.n_small)
      pr_tile_runs_small_kernel<<<grid_for(g.n_small), kBlock, 0, s>>>(
          pull_off, pull_dst, g.rows_small.data(), g.n_small, g.v_begin, c,
          out);
    if (g.n_mid)
      pr_tile_runs_wave_kernel<<<grid_for(g.n_mid * kWave), kBlock, 0, s>>>(
          pull_off, pull_dst, g.rows_mid.data(), g.n_mid, g.v_begin, c, out);
    if (g.n_large)
      pr_tile_runs_wave_kernel<<<std::min<uint64_t>(g.n_large, kMaxGrid),
                                 kBlock, 0, s>>>(
          pull_off, pull_dst, g.rows_large.data(), g.n_large, g.v_begin, c,
          out);
  };
  runs(cnt.data(), nullptr);
  g.pr_tile_off.resize(static_cast<size_t>(ntiles) + 1);
  exclusive_scan(cnt.data(), g.pr_tile_off.data(), ntiles, s, I.scan);
  g.pr_tiles.resize(ne ? ne : 1);
  // cursors = copy of offsets
  DeviceBuffer<unsigned long long> cur(ntiles + 1);
  HIP_CHECK(hipMemcpyAsync(cur.data(), g.pr_tile_off.data(),
                           (static_cast<size_t>(ntiles) + 1) * 8,
                           hipMemcpyDeviceToDevice, s));
  runs(cur.data(), g.pr_tiles.data());
  // host copies of the slice-boundary tile offsets for the phase split
  uint32_t t0 = (g.v_begin + (1u << kPrTileBits) - 1) >> kPrTileBits;
  uint32_t t1 = g.v_end >> kPrTileBits;
  if (t1 < t0) t1 = t0;
  if (t1 > ntiles) t1 = ntiles;
  uint64_t v[3] = {0, 0, 0};
  HIP_CHECK(hipMemcpyAsync(&v[0], g.pr_tile_off.data() + t0, 8,
                           hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipMemcpyAsync(&v[1], g.pr_tile_off.data() + t1, 8,
                           hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipMemcpyAsync(&v[2], g.pr_tile_off.data() + ntiles, 8,
                           hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipStreamSynchronize(s));
  g.pr_own_lo = v[0];
  g.pr_own_hi = v[1];
  g.pr_total = v[2];
  g.pr_ntiles = ntiles;
  g.pr_tiles_built = true;
}

// ---------------------------------------------------------------------------
// Materialize a bitmap frontier into a queue; returns the count.
// bm covers bits [0, nbits) (owned rows); q entries are global vids
// (bit_base = v_begin). Clears the bitmap.
// ---------------------------------------------------------------------------
static uint64_t compact_frontier(GpuContext::Impl& I, uint32_t* bm_words,
                                 size_t nbits, uint32_t bit_base, uint32_t* q,
                                 hipStream_t s) {
  size_t nwords = (nbits + 31) / 32;
  if (I.frontier_deg.size() < nwords)
    I.frontier_deg.resize(nwords + (nwords >> 2) + 64);
  if (I.frontier_off.size() < nwords + 1)
    I.frontier_off.resize(nwords + (nwords >> 2) + 65);
  popc_words_kernel<<<grid_for(nwords), kBlock, 0, s>>>(bm_words, nwords,
                                                        I.frontier_deg.data());
  uint64_t total = exclusive_scan(I.frontier_deg.data(),
                                  I.frontier_off.data(), nwords, s, I.scan);
  if (total)
    fill_frontier_kernel<<<grid_for(nwords), kBlock, 0, s>>>(
        bm_words, nwords, I.frontier_off.data(), bit_base, q);
  return total;
}

// ---------------------------------------------------------------------------
// Halo host-side flush: pack per-peer pairs, exchange counts over TCP,
// ncclSend/Recv payloads over xGMI. Returns #received pairs (in recvbuf).
// ---------------------------------------------------------------------------
template <typename T>
uint64_t halo_flush(GpuContext::Impl& I, TcpComm* comm, int rank, int world,
                    const T* state, uint64_t cap, hipStream_t s) {
  std::vector<unsigned long long> cnts(world);
  HIP_CHECK(hipMemcpyAsync(cnts.data(), I.halo_cnt.data(), world * 8,
                           hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipStreamSynchronize(s));  // the one inherent host sync: the
                                       // pack launch shapes need the counts
  uint64_t send_total = 0;
  for (int p = 0; p < world; ++p) send_total += cnts[p];
  size_t pair_sz = sizeof(HaloPair<T>);
  if (I.sendbuf.size() < send_total * pair_sz)
    I.sendbuf.resize(send_total * pair_sz + (1 << 20));
  // pack contiguously in peer order (async: runs while the TCP count
  // matrix goes around on the host below)
  std::vector<uint64_t> send_off(world + 1, 0);
  for (int p = 0; p < world; ++p)
    send_off[p + 1] = send_off[p] + cnts[p] * pair_sz;
  for (int p = 0; p < world; ++p) {
    if (!cnts[p]) continue;
    halo_pack_kernel<T><<<grid_for(cnts[p]), kBlock, 0, s>>>(
        I.halo_idx.data() + static_cast<uint64_t>(p) * cap, cnts[p], state,
        DevBitmap{I.halo_bm.data()},
        reinterpret_cast<HaloPair<T>*>(I.sendbuf.data() + send_off[p]));
  }
  I.halo_cnt.zero(s);
  // exchange the count matrix on the control plane
  std::vector<uint64_t> matrix(static_cast<size_t>(world) * world);
  std::vector<uint64_t> mine(cnts.begin(), cnts.end());
  comm->allgather(mine.data(), world * 8, matrix.data());
  uint64_t recv_total = 0;
  std::vector<uint64_t> recv_off(world + 1, 0);
  for (int p = 0; p < world; ++p) {
    uint64_t from_p = matrix[static_cast<size_t>(p) * world + rank];
    recv_off[p + 1] = recv_off[p] + from_p * pair_sz;
    recv_total += from_p;
  }
  if (I.recvbuf.size() < recv_total * pair_sz)
    I.recvbuf.resize(recv_total * pair_sz + (1 << 20));
  // payloads over xGMI on the comm stream, event-ordered after the packs
  // (reference gpu_message_manager.h:401-445 dual-stream design); no host
  // sync — the caller's halo_process launch on `s` waits on ev_comm.
  I.ev_pack.record(s);
  I.ev_pack.wait_on(I.comm_stream);
  I.dc.sendrecv(I.sendbuf.data(), send_off, I.recvbuf.data(), recv_off,
                I.comm_stream);
  I.ev_comm.record(I.comm_stream);
  I.ev_comm.wait_on(s);
  // self pairs (should be none — owned handled locally)
  return recv_total;
}

// ===========================================================================
// Mirror topology: per-peer lists of the remote vertices this rank's edges
// reference (reference mirror-info + BatchShuffle, edgecut_fragment_base.h
// :569+ / cuda batch_shuffle_message_manager.h:81-103). Built once per
// graph; per-round dense refreshes then ship ONLY referenced values,
// point-to-point — each peer pair on its own xGMI link — instead of
// allgathering whole slices around the ring (O(V·world) volume and
// per-link serialization). 4-byte element specialization (depth u32,
// PR contrib f32, CDLP label u32).
// ===========================================================================

__global__ void mark_dsts_kernel(const uint32_t* __restrict__ dst,
                                 uint64_t n, uint32_t v_begin,
                                 uint32_t v_end, DevBitmap bm) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < n; i += stride) {
    uint32_t v = dst[i];
    if (v < v_begin || v >= v_end) (void)bm.set_once(v);
  }
}

// out[t] = lower_bound(idx[0..n), t*slice) for t in [0, world]
__global__ void slice_bounds_kernel(const uint32_t* __restrict__ idx,
                                    uint64_t n, uint32_t slice, int world,
                                    uint64_t* __restrict__ out) {
  int t = threadIdx.x;
  if (t > world) return;
  uint64_t key = static_cast<uint64_t>(t) * slice;
  uint64_t lo = 0, hi = n;
  while (lo < hi) {
    uint64_t m = (lo + hi) >> 1;
    if (idx[m] < key) lo = m + 1;
    else hi = m;
  }
  out[t] = lo;
}

__global__ void gather4_kernel(const uint32_t* __restrict__ idx, uint64_t n,
                               const uint32_t* __restrict__ state,
                               uint32_t* __restrict__ out) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < n; i += stride)
    out[i] = state[idx[i]];
}

__global__ void scatter4_kernel(const uint32_t* __restrict__ idx, uint64_t n,
                                const uint32_t* __restrict__ in,
                                uint32_t* __restrict__ state) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < n; i += stride)
    state[idx[i]] = in[i];
}

// Build per-peer request lists from a referenced bitmap (consumed/cleared).
// recv_idx: ascending global ids this rank wants values for, regioned per
// owner; send_idx: the ids each peer wants from this rank's slice.
// Returns the recv total.
static uint64_t build_ref_lists(GpuContext::Impl& I, TcpComm* comm,
                                DeviceBuffer<uint32_t>& bm, size_t nwords,
                                uint32_t slice, int rank, int world,
                                DeviceBuffer<uint32_t>& recv_idx,
                                std::vector<uint64_t>& recv_off,
                                DeviceBuffer<uint32_t>& send_idx,
                                std::vector<uint64_t>& send_off,
                                hipStream_t s) {
  // compact ascending (popc + scan + fill, cleared as it fills)
  if (I.frontier_deg.size() < nwords)
    I.frontier_deg.resize(nwords + (nwords >> 2) + 64);
  if (I.frontier_off.size() < nwords + 1)
    I.frontier_off.resize(nwords + (nwords >> 2) + 65);
  popc_words_kernel<<<grid_for(nwords), kBlock, 0, s>>>(bm.data(), nwords,
                                                        I.frontier_deg.data());
  uint64_t total = exclusive_scan(I.frontier_deg.data(),
                                  I.frontier_off.data(), nwords, s, I.scan);
  recv_idx.resize(total ? total : 1);
  if (total)
    fill_frontier_kernel<<<grid_for(nwords), kBlock, 0, s>>>(
        bm.data(), nwords, I.frontier_off.data(), 0, recv_idx.data());
  // per-peer boundaries (vid-ascending list, slice-contiguous regions)
  DeviceBuffer<uint64_t> d_bounds(world + 1);
  slice_bounds_kernel<<<1, world + 1, 0, s>>>(recv_idx.data(), total, slice,
                                              world, d_bounds.data());
  recv_off = d_bounds.download(s);
  // counts matrix on the control plane -> send-side region sizes
  std::vector<uint64_t> mine(world);
  for (int p = 0; p < world; ++p) mine[p] = recv_off[p + 1] - recv_off[p];
  std::vector<uint64_t> matrix(static_cast<size_t>(world) * world);
  comm->allgather(mine.data(), world * 8, matrix.data());
  send_off.assign(world + 1, 0);
  for (int q = 0; q < world; ++q)
    send_off[q + 1] =
        send_off[q] +
        (q == rank ? 0 : matrix[static_cast<size_t>(q) * world + rank]);
  uint64_t total_send = send_off[world];
  send_idx.resize(total_send ? total_send : 1);
  // ship each peer its request list; receive who wants what from us
  std::vector<uint64_t> sb(world + 1), rb(world + 1);
  for (int p = 0; p <= world; ++p) {
    sb[p] = recv_off[p] * 4;    // my requests, region p -> peer p
    rb[p] = send_off[p] * 4;    // peer q's requests for my slice
  }
  I.dc.sendrecv(reinterpret_cast<const uint8_t*>(recv_idx.data()), sb,
                reinterpret_cast<uint8_t*>(send_idx.data()), rb, s);
  HIP_CHECK(hipStreamSynchronize(s));
  return total;
}

static void ensure_mirrors(GpuContext::Impl& I, TcpComm* comm,
                           DeviceGraph& g, int rank, int world,
                           hipStream_t s) {
  if (g.mirrors_built || world <= 1) return;
  uint32_t nv_pad = padded_nv(g, world);
  uint32_t slice = g.seg_host.size() > 1 ? g.seg_host[1] - g.seg_host[0]
                                         : g.nv_global;
  size_t nwords = (static_cast<size_t>(nv_pad) + 31) / 32;
  DeviceBuffer<uint32_t> bm(nwords);
  bm.zero(s);
  if (g.local_edges)
    mark_dsts_kernel<<<grid_for(g.local_edges), kBlock, 0, s>>>(
        g.oe_dst.data(), g.local_edges, g.v_begin, g.v_end, DevBitmap{bm.data()});
  if (g.has_in && g.ie_dst.size())
    mark_dsts_kernel<<<grid_for(g.ie_dst.size()), kBlock, 0, s>>>(
        g.ie_dst.data(), g.ie_dst.size(), g.v_begin, g.v_end,
        DevBitmap{bm.data()});
  uint64_t total =
      build_ref_lists(I, comm, bm, nwords, slice, rank, world,
                      g.mr_recv_idx, g.mr_recv_off, g.mr_send_idx,
                      g.mr_send_off, s);
  uint64_t total_send = g.mr_send_off[world];
  g.mr_sendbuf.resize(total_send * 4 + 4);
  g.mr_recvbuf.resize(total * 4 + 4);
  g.mirrors_built = true;
  if (getenv("GRAPEHIP_DEBUG"))
    fprintf(stderr,
            "[mirror] rank %d: recv %lu refs, send %lu (slice %u)\n", rank,
            (unsigned long)total, (unsigned long)total_send, slice);
}

// Refresh referenced remote entries of a 4-byte-element state array over
// arbitrary request lists. begin() packs on the compute stream and
// launches the exchange on the comm stream; independent local work may
// run between begin and end — end() orders the scatter after payloads
// land.
static void ref_sync_begin(GpuContext::Impl& I,
                           const DeviceBuffer<uint32_t>& send_idx,
                           const std::vector<uint64_t>& send_off,
                           const std::vector<uint64_t>& recv_off,
                           DeviceBuffer<uint8_t>& sendbuf,
                           DeviceBuffer<uint8_t>& recvbuf,
                           const void* state4, hipStream_t s) {
  uint64_t ns = send_off.back();
  if (ns)
    gather4_kernel<<<grid_for(ns), kBlock, 0, s>>>(
        send_idx.data(), ns, static_cast<const uint32_t*>(state4),
        reinterpret_cast<uint32_t*>(sendbuf.data()));
  I.ev_pack.record(s);
  I.ev_pack.wait_on(I.comm_stream);
  std::vector<uint64_t> sb(send_off.size()), rb(recv_off.size());
  for (size_t p = 0; p < sb.size(); ++p) sb[p] = send_off[p] * 4;
  for (size_t p = 0; p < rb.size(); ++p) rb[p] = recv_off[p] * 4;
  I.dc.sendrecv(sendbuf.data(), sb, recvbuf.data(), rb, I.comm_stream);
  I.ev_comm.record(I.comm_stream);
}

static void ref_sync_end(GpuContext::Impl& I,
                         const DeviceBuffer<uint32_t>& recv_idx,
                         const std::vector<uint64_t>& recv_off,
                         const DeviceBuffer<uint8_t>& recvbuf, void* state4,
                         hipStream_t s) {
  I.ev_comm.wait_on(s);
  uint64_t nr = recv_off.back();
  if (nr)
    scatter4_kernel<<<grid_for(nr), kBlock, 0, s>>>(
        recv_idx.data(), nr,
        reinterpret_cast<const uint32_t*>(recvbuf.data()),
        static_cast<uint32_t*>(state4));
}

static void mirror_sync_begin(GpuContext::Impl& I, DeviceGraph& g,
                              const void* state4, hipStream_t s) {
  ref_sync_begin(I, g.mr_send_idx, g.mr_send_off, g.mr_recv_off,
                 g.mr_sendbuf, g.mr_recvbuf, state4, s);
}

static void mirror_sync_end(GpuContext::Impl& I, DeviceGraph& g,
                            void* state4, hipStream_t s) {
  ref_sync_end(I, g.mr_recv_idx, g.mr_recv_off, g.mr_recvbuf, state4, s);
}

// Changed-gated sparse refresh: ship only (id, value) pairs whose OWNED
// row is marked in `chg` (bit index = id - v_begin). Converging apps
// (CDLP late iterations) change few labels — the dense mirror refresh
// would re-ship every referenced value regardless. Pair order within a
// peer is irrelevant (receiver scatters by id), so packing appends with
// plain atomic cursors into capacity regions; per-round counts ride the
// control plane.
__global__ void mirror_pack_sparse_kernel(
    const uint32_t* __restrict__ send_idx, uint64_t b, uint64_t e,
    const uint32_t* __restrict__ chg_words, uint32_t v_begin,
    const uint32_t* __restrict__ state, unsigned long long* __restrict__ cnt,
    unsigned long long* __restrict__ out) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = b + blockIdx.x * blockDim.x + threadIdx.x; i < e;
       i += stride) {
    uint32_t v = send_idx[i];
    uint32_t bit = v - v_begin;
    if ((chg_words[bit >> 5] >> (bit & 31)) & 1u) {
      unsigned long long pos = atomicAdd(cnt, 1ull);
      out[b + pos] = (static_cast<unsigned long long>(v) << 32) | state[v];
    }
  }
}

__global__ void mirror_scatter_pairs_kernel(
    const unsigned long long* __restrict__ pairs, uint64_t n,
    uint32_t* __restrict__ state) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < n; i += stride)
    state[pairs[i] >> 32] = static_cast<uint32_t>(pairs[i]);
}

static void mirror_sync_changed(GpuContext::Impl& I, TcpComm* comm,
                                DeviceGraph& g, void* state4,
                                const uint32_t* chg_words, int rank,
                                int world, hipStream_t s) {
  uint64_t ns = g.mr_send_off.back();
  uint64_t nr = g.mr_recv_off.back();
  // pair staging reuses the mirror buffers when large enough (8 B/entry;
  // recvbuf doubles as the compaction bounce, so size it for both sides)
  uint64_t big = (ns > nr ? ns : nr) * 8 + 8;
  if (g.mr_sendbuf.size() < ns * 8 + 8) g.mr_sendbuf.resize(ns * 8 + 8);
  if (g.mr_recvbuf.size() < big) g.mr_recvbuf.resize(big);
  DeviceBuffer<unsigned long long> cnts(world);
  cnts.zero(s);
  auto* pairs = reinterpret_cast<unsigned long long*>(g.mr_sendbuf.data());
  for (int q = 0; q < world; ++q) {
    uint64_t b = g.mr_send_off[q], e = g.mr_send_off[q + 1];
    if (e > b)
      mirror_pack_sparse_kernel<<<grid_for(e - b), kBlock, 0, s>>>(
          g.mr_send_idx.data(), b, e, chg_words, g.v_begin,
          static_cast<const uint32_t*>(state4), cnts.data() + q, pairs);
  }
  auto hc = cnts.download(s);
  std::vector<uint64_t> mine(world);
  for (int q = 0; q < world; ++q) mine[q] = hc[q];
  std::vector<uint64_t> matrix(static_cast<size_t>(world) * world);
  comm->allgather(mine.data(), world * 8, matrix.data());
  std::vector<uint64_t> sb(world + 1), rb(world + 1);
  sb[0] = 0;
  rb[0] = 0;
  for (int q = 0; q < world; ++q) sb[q + 1] = sb[q] + mine[q] * 8;
  uint64_t recv_pairs = 0;
  for (int p = 0; p < world; ++p) {
    uint64_t from_p =
        p == rank ? 0 : matrix[static_cast<size_t>(p) * world + rank];
    rb[p + 1] = rb[p] + from_p * 8;
    recv_pairs += from_p;
  }
  // compact the capacity regions into contiguous send order: copy each
  // peer's [b*8, b*8 + cnt*8) down — regions are already at b*8; the
  // contiguous layout needs offsets sb. D2D per peer (async, small).
  for (int q = world - 1; q >= 0; --q) {
    if (!mine[q] || g.mr_send_off[q] * 8 == sb[q]) continue;
    HIP_CHECK(hipMemcpyAsync(
        g.mr_recvbuf.data() + sb[q],
        g.mr_sendbuf.data() + g.mr_send_off[q] * 8, mine[q] * 8,
        hipMemcpyDeviceToDevice, s));
  }
  // regions that moved were staged into recvbuf — move back contiguously
  for (int q = world - 1; q >= 0; --q) {
    if (!mine[q] || g.mr_send_off[q] * 8 == sb[q]) continue;
    HIP_CHECK(hipMemcpyAsync(g.mr_sendbuf.data() + sb[q],
                             g.mr_recvbuf.data() + sb[q], mine[q] * 8,
                             hipMemcpyDeviceToDevice, s));
  }
  I.ev_pack.record(s);
  I.ev_pack.wait_on(I.comm_stream);
  I.dc.sendrecv(g.mr_sendbuf.data(), sb, g.mr_recvbuf.data(), rb,
                I.comm_stream);
  I.ev_comm.record(I.comm_stream);
  I.ev_comm.wait_on(s);
  if (recv_pairs)
    mirror_scatter_pairs_kernel<<<grid_for(recv_pairs), kBlock, 0, s>>>(
        reinterpret_cast<const unsigned long long*>(g.mr_recvbuf.data()),
        recv_pairs, static_cast<uint32_t*>(state4));
}

// old-id -> renumbered-id for API source arguments
static uint32_t map_source(DeviceGraph& g, int64_t source, hipStream_t s) {
  uint32_t v = static_cast<uint32_t>(source);
  if (!g.permuted || v >= g.perm.size()) return v;
  uint32_t out = 0;
  HIP_CHECK(hipMemcpyAsync(&out, g.perm.data() + v, 4,
                           hipMemcpyDeviceToHost, s));
  HIP_CHECK(hipStreamSynchronize(s));
  return out;
}

// canonical WCC labels under renumbering: component label = min OLD id
__global__ void wcc_minold_kernel(const uint32_t* __restrict__ parent,
                                  const uint32_t* __restrict__ inv,
                                  uint32_t n, uint32_t* __restrict__ m) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t v = blockIdx.x * blockDim.x + threadIdx.x; v < n;
       v += stride)
    atomicMin(&m[parent[v]], inv[v]);
}

__global__ void wcc_relabel_kernel(const uint32_t* __restrict__ parent,
                                   const uint32_t* __restrict__ m,
                                   uint32_t v_begin, uint32_t n,
                                   uint32_t* __restrict__ out) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    out[i] = m[parent[v_begin + i]];
}

// generic frontier expansion helper: scan degrees then CM-expand
template <typename Op>
void expand_frontier(GpuContext::Impl& I, const DevGraphView& view,
                     const uint32_t* q, uint32_t qn, Op op, hipStream_t s) {
  if (qn == 0) return;
  if (I.frontier_deg.size() < qn) I.frontier_deg.resize(qn + (qn >> 2) + 64);
  if (I.frontier_off.size() < qn + 1)
    I.frontier_off.resize(qn + (qn >> 2) + 65);
  gather_deg_kernel<<<grid_for(qn), kBlock, 0, s>>>(view, q, qn,
                                                    I.frontier_deg.data());
  uint64_t total = exclusive_scan(I.frontier_deg.data(),
                                  I.frontier_off.data(), qn, s, I.scan);
  // LB strategy (reference --lb flag, default cta/cm): cm = LDS-staged
  // owner search; wm = wave-granular register-staged CM; strict =
  // perfectly edge-balanced; none = thread-per-row
  static const char* lb = getenv("GRAPEHIP_LB");
  if (lb && lb[0] == 's') {
    expand_strict_frontier<Op><<<grid_for(total), kBlock, 0, s>>>(
        view, q, qn, I.frontier_off.data(), total, op);
  } else if (lb && lb[0] == 'n') {
    expand_none_frontier<Op><<<grid_for(qn), kBlock, 0, s>>>(view, q, qn,
                                                             op);
  } else if (lb && lb[0] == 'w') {
    expand_wm_frontier<Op><<<grid_for(qn), kBlock, 0, s>>>(view, q, qn,
                                                           op);
  } else {
    int nchunks = static_cast<int>((qn + kBlock - 1) / kBlock);
    expand_cm_frontier<Op><<<std::min(nchunks, kMaxGrid), kBlock, 0, s>>>(
        view, q, qn, I.frontier_off.data(), op);
  }
}

// ---------------------------------------------------------------------------
// BFS
// ---------------------------------------------------------------------------
GpuRunResult GpuContext::bfs(DeviceGraph& g, int64_t source,
                             bool fetch) {
  RangeMarker _mk("grapehip::bfs");
  auto& I = *impl_;
  hipStream_t s = I.compute;
  DevGraphView view = make_view(g, rank_, world_);
  uint32_t nv_pad = padded_nv(g, world_);
  uint32_t owned = g.owned();
  uint64_t cap = view.slice;
  bool multi = world_ > 1;

  const bool pull_capable = !g.directed || g.has_in;
  if (pull_capable) ensure_buckets(g, s);
  if (multi && pull_capable) ensure_mirrors(I, comm_, g, rank_, world_, s);
  const uint64_t* pull_off = !g.directed ? g.oe_off.data() : g.ie_off.data();
  const uint32_t* pull_dst = !g.directed ? g.oe_dst.data() : g.ie_dst.data();

  DeviceBuffer<uint32_t> depth(nv_pad);
  DeviceBuffer<uint32_t> q(owned + 64);
  size_t bm_words = (static_cast<size_t>(owned) + 31) / 32 + 1;
  DeviceBuffer<uint32_t> next_bm(bm_words);
  DeviceBuffer<uint32_t> fdeg;   // frontier degrees (separate from compaction
  DeviceBuffer<uint64_t> foff;   //  scratch so both stay live in one round)
  if (multi) {
    I.halo_idx.resize(static_cast<uint64_t>(world_) * cap);
    I.halo_cnt.resize(world_);
    I.halo_bm.resize((static_cast<uint64_t>(nv_pad) + 31) / 32);
  }

  if (comm_) comm_->barrier();
  HIP_CHECK(hipDeviceSynchronize());
  double t0 = wall_s();
  const uint64_t b_p2p0 = I.dc.bytes_p2p, b_coll0 = I.dc.bytes_coll;

  depth.fill_bytes(0xFF, s);
  next_bm.zero(s);
  if (multi) {
    I.halo_cnt.zero(s);
    I.halo_bm.zero(s);
  }
  uint32_t src = map_source(g, source, s);
  if (src >= g.v_begin && src < g.v_end)
    bfs_seed_kernel<<<1, 1, 0, s>>>(depth.data(), src, g.v_begin,
                                    next_bm.data());
  uint32_t qn =
      static_cast<uint32_t>(compact_frontier(I, next_bm.data(), owned,
                                             g.v_begin, q.data(), s));
  uint64_t global_curr = multi ? comm_->allreduce_sum(qn) : qn;
  uint32_t level = 0;
  int rounds = 0;
  // Beamer-style switch: pull when the frontier's out-edges exceed a
  // fraction of the stored edges (cuda/bfs/bfs.h heuristic, retuned).
  static const int pull_div = [] {
    const char* e = getenv("GRAPEHIP_BFS_PULL_DIV");
    return e ? atoi(e) : 16;
  }();
  const uint64_t pull_edge_threshold = g.total_edges / pull_div;
  while (global_curr > 0) {
    bool use_pull = false;
    uint64_t fedges = 0;
    if (qn) {
      if (fdeg.size() < qn) fdeg.resize(qn + (qn >> 2) + 64);
      if (foff.size() < qn + 1) foff.resize(qn + (qn >> 2) + 65);
      gather_deg_kernel<<<grid_for(qn), kBlock, 0, s>>>(view, q.data(), qn,
                                                        fdeg.data());
      fedges = exclusive_scan(fdeg.data(), foff.data(), qn, s, I.scan);
    }
    if (pull_capable) {
      uint64_t gedges = multi ? comm_->allreduce_sum(fedges) : fedges;
      use_pull = gedges > pull_edge_threshold;
    }
    if (use_pull) {
      if (multi) {
        // refresh referenced remote depths only (vs allgathering every
        // slice): the pull scan reads depth[dst] for local dsts alone
        mirror_sync_begin(I, g, depth.data(), s);
        mirror_sync_end(I, g, depth.data(), s);
      }
      DevBitmap nb{next_bm.data()};
      if (g.n_small)
        bfs_pull_small_kernel<<<grid_for(g.n_small), kBlock, 0, s>>>(
            pull_off, pull_dst, g.rows_small.data(), g.n_small, g.v_begin,
            depth.data(), level, nb);
      if (g.n_mid)
        bfs_pull_mid_kernel<<<grid_for(g.n_mid * kWave), kBlock, 0, s>>>(
            pull_off, pull_dst, g.rows_mid.data(), g.n_mid, g.v_begin,
            depth.data(), level, nb);
      if (g.n_large)
        bfs_pull_large_kernel<<<static_cast<int>(std::min<uint64_t>(
                                    std::max<uint64_t>(g.n_large, 1),
                                    kMaxGrid)),
                                kBlock, 0, s>>>(
            pull_off, pull_dst, g.rows_large.data(), g.n_large, g.v_begin,
            depth.data(), level, nb);
    } else {
      BfsOp op{depth.data(), level + 1, DevBitmap{next_bm.data()}, g.v_begin,
               g.v_end, multi,
               DevHalo{I.halo_idx.data(), I.halo_cnt.data(),
                       DevBitmap{I.halo_bm.data()}, cap, view.slice, world_}};
      if (qn) {
        int nchunks = static_cast<int>((qn + kBlock - 1) / kBlock);
        expand_cm_frontier<BfsOp><<<std::min(nchunks, kMaxGrid), kBlock, 0,
                                    s>>>(view, q.data(), qn, foff.data(), op);
      }
      if (multi) {
        uint64_t nrecv = halo_flush<uint32_t>(I, comm_, rank_, world_,
                                              depth.data(), cap, s);
        if (nrecv)
          halo_process_kernel<uint32_t, BfsRecvOp>
              <<<grid_for(nrecv), kBlock, 0, s>>>(
                  reinterpret_cast<HaloPair<uint32_t>*>(I.recvbuf.data()),
                  nrecv,
                  BfsRecvOp{depth.data(), DevBitmap{next_bm.data()},
                            g.v_begin});
      }
    }
    qn = static_cast<uint32_t>(compact_frontier(I, next_bm.data(), owned,
                                                g.v_begin, q.data(), s));
    global_curr = multi ? comm_->allreduce_sum(qn) : qn;
    if (getenv("GRAPEHIP_DEBUG"))
      fprintf(stderr, "[bfs] level=%u mode=%s fedges=%lu next=%u t=%.1fms\n",
              level, use_pull ? "pull" : "push", (unsigned long)fedges, qn,
              (wall_s() - t0) * 1e3);
    ++level;
    ++rounds;
  }
  HIP_CHECK(hipDeviceSynchronize());
  if (comm_) comm_->barrier();
  double t1 = wall_s();

  GpuRunResult res;
  res.bytes_p2p = I.dc.bytes_p2p - b_p2p0;
  res.bytes_coll = I.dc.bytes_coll - b_coll0;
  res.rounds = rounds;
  res.seconds = comm_ ? comm_->allreduce_max_double(t1 - t0) : (t1 - t0);
  res.traversed_edges = g.input_edges;
  if (fetch) {
    std::vector<uint32_t> d32(owned);
    HIP_CHECK(hipMemcpyAsync(d32.data(), depth.data() + g.v_begin, owned * 4,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    res.i64.resize(owned);
    for (uint32_t i = 0; i < owned; ++i)
      res.i64[i] = d32[i] == 0xFFFFFFFFu
                       ? std::numeric_limits<int64_t>::max()
                       : static_cast<int64_t>(d32[i]);
  }
  return res;
}

// ---------------------------------------------------------------------------
// SSSP (near-far delta stepping; cuda/sssp/sssp.h parity)
// ---------------------------------------------------------------------------
GpuRunResult GpuContext::sssp(DeviceGraph& g, int64_t source,
                              float delta, bool fetch) {
  RangeMarker _mk("grapehip::sssp");
  auto& I = *impl_;
  hipStream_t s = I.compute;
  DevGraphView view = make_view(g, rank_, world_);
  uint32_t nv_pad = padded_nv(g, world_);
  uint32_t owned = g.owned();
  uint64_t cap = view.slice;
  bool multi = world_ > 1;

  DeviceBuffer<float> dist(nv_pad);
  DeviceBuffer<uint32_t> q0(owned + 64), q1(owned + 64);   // near dbl-buf
  DeviceBuffer<uint32_t> qf0(owned + 64), qf1(owned + 64);  // far dbl-buf
  DeviceBuffer<unsigned long long> fcnt(2);  // far queue counters
  size_t own_words = (static_cast<size_t>(owned) + 31) / 32 + 1;
  uint64_t glob_words = (static_cast<uint64_t>(nv_pad) + 31) / 32;
  DeviceBuffer<uint32_t> near_bm(own_words), far_bm(glob_words);
  DeviceBuffer<uint32_t> fdeg;
  DeviceBuffer<uint64_t> foff;
  if (multi) {
    I.halo_idx.resize(static_cast<uint64_t>(world_) * cap);
    I.halo_cnt.resize(world_);
    I.halo_bm.resize(glob_words);
  }
  if (delta <= 0) {
    // sweep across LiveJournal/orkut/datagen shapes put the optimum on a
    // wide plateau around 16x the mean edge weight (profiles/ r01 sweep);
    // the degree-scaled Davidson heuristic was 2x off on dense graphs.
    double mean_w = 1.0;
    if (g.weighted && g.local_edges) {
      size_t n_sample = std::min<uint64_t>(g.local_edges, 1u << 20);
      std::vector<float> h(n_sample);
      HIP_CHECK(hipMemcpyAsync(h.data(), g.oe_w.data(), n_sample * 4,
                               hipMemcpyDeviceToHost, s));
      HIP_CHECK(hipStreamSynchronize(s));
      double acc = 0;
      for (float w : h) acc += w;
      mean_w = acc / n_sample;
    }
    if (comm_ && world_ > 1)
      mean_w = comm_->allreduce_max_double(mean_w);
    delta = static_cast<float>(16.0 * mean_w);
  }

  if (comm_) comm_->barrier();
  HIP_CHECK(hipDeviceSynchronize());
  double t0 = wall_s();
  const uint64_t b_p2p0 = I.dc.bytes_p2p, b_coll0 = I.dc.bytes_coll;

  fill(dist.data(), std::numeric_limits<float>::max(), nv_pad, s);
  fcnt.zero(s);
  near_bm.zero(s);
  far_bm.zero(s);
  if (multi) {
    I.halo_cnt.zero(s);
    I.halo_bm.zero(s);
  }
  uint32_t src = map_source(g, source, s);
  if (src >= g.v_begin && src < g.v_end)
    sssp_seed_kernel<<<1, 1, 0, s>>>(dist.data(), src, g.v_begin,
                                     near_bm.data());
  uint32_t* nearq[2] = {q0.data(), q1.data()};
  uint32_t* farq[2] = {qf0.data(), qf1.data()};
  int ncur = 0, fcur = 0;
  float prio = 0.0f;
  int rounds = 0;
  unsigned long long zero = 0;
  std::vector<unsigned long long> hfc(2);
  uint32_t qn = static_cast<uint32_t>(compact_frontier(
      I, near_bm.data(), owned, g.v_begin, nearq[ncur], s));
  uint64_t g_near = multi ? comm_->allreduce_sum(qn) : qn;
  for (;;) {
    float prio_hi = prio + delta;
    if (qn) {
      if (fdeg.size() < qn) fdeg.resize(qn + (qn >> 2) + 64);
      if (foff.size() < qn + 1) foff.resize(qn + (qn >> 2) + 65);
      gather_deg_kernel<<<grid_for(qn), kBlock, 0, s>>>(view, nearq[ncur], qn,
                                                        fdeg.data());
      exclusive_scan(fdeg.data(), foff.data(), qn, s, I.scan);
      SsspOp op{dist.data(), prio_hi, DevBitmap{near_bm.data()},
                DevQueue{farq[fcur], fcnt.data() + fcur},
                DevBitmap{far_bm.data()}, g.v_begin, g.v_end, multi,
                DevHalo{I.halo_idx.data(), I.halo_cnt.data(),
                        DevBitmap{I.halo_bm.data()}, cap, view.slice,
                        world_}};
      int nchunks = static_cast<int>((qn + kBlock - 1) / kBlock);
      expand_cm_frontier<SsspOp><<<std::min(nchunks, kMaxGrid), kBlock, 0,
                                   s>>>(view, nearq[ncur], qn, foff.data(),
                                        op);
    }
    if (multi) {
      uint64_t nrecv =
          halo_flush<float>(I, comm_, rank_, world_, dist.data(), cap, s);
      if (nrecv)
        halo_process_kernel<float, SsspRecvOp>
            <<<grid_for(nrecv), kBlock, 0, s>>>(
                reinterpret_cast<HaloPair<float>*>(I.recvbuf.data()), nrecv,
                SsspRecvOp{dist.data(), prio_hi, DevBitmap{near_bm.data()},
                           DevQueue{farq[fcur], fcnt.data() + fcur},
                           DevBitmap{far_bm.data()}, g.v_begin});
    }
    ++rounds;
    uint32_t next_qn = static_cast<uint32_t>(compact_frontier(
        I, near_bm.data(), owned, g.v_begin, nearq[1 - ncur], s));
    g_near = multi ? comm_->allreduce_sum(next_qn) : next_qn;
    if (getenv("GRAPEHIP_DEBUG"))
      fprintf(stderr, "[sssp] round=%d prio=%.0f in=%u next=%u t=%.1fms\n",
              rounds, prio, qn, next_qn, (wall_s() - t0) * 1e3);
    ncur = 1 - ncur;
    qn = next_qn;
    if (g_near > 0) continue;
    // near exhausted globally: advance priority, repartition far buckets
    HIP_CHECK(hipMemcpyAsync(hfc.data(), fcnt.data(), 16,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    uint64_t far_n = hfc[fcur];
    uint64_t g_far = multi ? comm_->allreduce_sum(far_n) : far_n;
    if (g_far == 0) break;
    while (g_near == 0 && g_far > 0) {
      prio += delta;
      float new_hi = prio + delta;
      if (far_n)
        sssp_repart_kernel<<<grid_for(far_n), kBlock, 0, s>>>(
            farq[fcur], static_cast<uint32_t>(far_n), dist.data(), new_hi,
            DevBitmap{near_bm.data()}, g.v_begin,
            DevQueue{farq[1 - fcur], fcnt.data() + (1 - fcur)},
            DevBitmap{far_bm.data()});
      HIP_CHECK(hipMemcpyAsync(fcnt.data() + fcur, &zero, 8,
                               hipMemcpyHostToDevice, s));
      qn = static_cast<uint32_t>(compact_frontier(
          I, near_bm.data(), owned, g.v_begin, nearq[ncur], s));
      g_near = multi ? comm_->allreduce_sum(qn) : qn;
      HIP_CHECK(hipMemcpyAsync(hfc.data(), fcnt.data(), 16,
                               hipMemcpyDeviceToHost, s));
      HIP_CHECK(hipStreamSynchronize(s));
      fcur = 1 - fcur;
      far_n = hfc[fcur];
      g_far = multi ? comm_->allreduce_sum(far_n) : far_n;
    }
    if (g_near == 0 && g_far == 0) break;
  }
  HIP_CHECK(hipDeviceSynchronize());
  if (comm_) comm_->barrier();
  double t1 = wall_s();

  GpuRunResult res;
  res.bytes_p2p = I.dc.bytes_p2p - b_p2p0;
  res.bytes_coll = I.dc.bytes_coll - b_coll0;
  res.rounds = rounds;
  res.seconds = comm_ ? comm_->allreduce_max_double(t1 - t0) : (t1 - t0);
  res.traversed_edges = g.input_edges;
  if (fetch) {
    std::vector<float> d32(owned);
    HIP_CHECK(hipMemcpyAsync(d32.data(), dist.data() + g.v_begin, owned * 4,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    res.f64.resize(owned);
    for (uint32_t i = 0; i < owned; ++i)
      res.f64[i] = d32[i] >= std::numeric_limits<float>::max()
                       ? std::numeric_limits<double>::max()
                       : static_cast<double>(d32[i]);
  }
  return res;
}

// ---------------------------------------------------------------------------
// PageRank (push + fp64 hw atomics; reduce-scatter/allgather over xGMI)
// ---------------------------------------------------------------------------
GpuRunResult GpuContext::pagerank(DeviceGraph& g, double damping,
                                  int iters, double tol, bool fetch) {
  RangeMarker _mk("grapehip::pagerank");
  auto& I = *impl_;
  hipStream_t s = I.compute;
  uint32_t nv_pad = padded_nv(g, world_);
  uint32_t owned = g.owned();
  uint32_t slice = g.seg_host.size() > 1 ? g.seg_host[1] - g.seg_host[0]
                                         : g.nv_global;
  bool multi = world_ > 1;
  const double N = static_cast<double>(g.nv_real);

  // pull path: undirected symmetric storage (out-CSR == neighborhood) or
  // directed with an in-CSR. Fallback: fp64-atomic push.
  const bool pull = !g.directed || g.has_in;
  const uint64_t* pull_off =
      (!g.directed ? g.oe_off.data() : g.ie_off.data());
  const uint32_t* pull_dst =
      (!g.directed ? g.oe_dst.data() : g.ie_dst.data());
  if (pull) ensure_buckets(g, s);
  if (pull && multi) ensure_mirrors(I, comm_, g, rank_, world_, s);
  // dst-tiled pull stream: auto-selected by density. On sparse shapes
  // (datagen mean degree ~5) the short per-src runs mean one fp64 L2
  // atomic per edge (~40 G/s cap) and the row pull wins ~3x; on dense
  // shapes (orkut mean degree ~142) runs amortize the atomics and the
  // tiled stream wins ~8% (same-box A/B: 46.3 vs 50.3 ms).
  // GRAPEHIP_PR_TILED={0,1} overrides.
  uint64_t pull_edges = !g.directed ? g.oe_dst.size() : g.ie_dst.size();
  double mean_pull_deg =
      g.owned_real ? static_cast<double>(pull_edges) / g.owned_real : 0.0;
  const char* te = getenv("GRAPEHIP_PR_TILED");
  const bool want_tiled =
      pull && (te ? atoi(te) != 0 : mean_pull_deg >= 64.0);
  if (want_tiled) ensure_pr_tiles(g, *impl_, s);
  const bool tiled = want_tiled && g.pr_ntiles > 0;

  // working set lives on the DeviceGraph: the cached hipGraph bakes these
  // pointers (locals would dangle across calls — GPU fault under realloc)
  if (g.pr_rank.size() < nv_pad) {
    g.pr_rank.resize(nv_pad);
    g.pr_acc.resize(nv_pad);
    g.pr_contrib.resize(nv_pad);
    g.pr_dangling.resize(1);
  }
  DeviceBuffer<double>& rank_arr = g.pr_rank;
  DeviceBuffer<double>& acc = g.pr_acc;
  DeviceBuffer<float>& contrib = g.pr_contrib;
  DeviceBuffer<double>& d_dangling = g.pr_dangling;
  if (d_dangling.size() < 2) d_dangling.resize(2);  // [0]=cur, [1]=next
  DeviceBuffer<double> d_l1(tol > 0 ? 1 : 0);
  DevGraphView view = make_view(g, rank_, world_);

  if (comm_) comm_->barrier();
  HIP_CHECK(hipDeviceSynchronize());
  double t0 = wall_s();
  const uint64_t b_p2p0 = I.dc.bytes_p2p, b_coll0 = I.dc.bytes_coll;

  fill(rank_arr.data(), 1.0 / N, nv_pad, s);
  int rounds = 0;
  const double inv_n = 1.0 / N;

  // phase-dispatched pull sweep (PHASE 0 whole row / 1 owned span / 2 rest)
  auto pulls = [&](int phase) {
    auto launch = [&](auto tag) {
      constexpr int P = decltype(tag)::value;
      if (g.n_small)
        pr_pull_small_cm_kernel<P><<<grid_for(g.n_small), kBlock, 0, s>>>(
            pull_off, pull_dst, contrib.data(), g.rows_small.data(),
            g.n_small, g.v_begin, g.v_begin, g.v_end, acc.data());
      if (g.n_mid)
        pr_pull_mid_kernel<P><<<grid_for(g.n_mid * kWave), kBlock, 0, s>>>(
            pull_off, pull_dst, contrib.data(), g.rows_mid.data(), g.n_mid,
            g.v_begin, g.v_begin, g.v_end, acc.data());
      if (g.n_large)
        pr_pull_large_kernel<P><<<std::min<uint64_t>(g.n_large, kMaxGrid),
                                  kBlock, 0, s>>>(
            pull_off, pull_dst, contrib.data(), g.rows_large.data(),
            g.n_large, g.v_begin, g.v_begin, g.v_end, acc.data());
    };
    if (phase == 0) launch(std::integral_constant<int, 0>{});
    else if (phase == 1) launch(std::integral_constant<int, 1>{});
    else launch(std::integral_constant<int, 2>{});
  };
  // fold the global dangling sum on the host via the pinned scalar
  // (the D2H was enqueued earlier; ev_scal gates only that copy, so the
  // GPU keeps crunching whatever was launched after it)
  auto fold_dangling = [&]() {
    I.ev_scal.sync();
    std::vector<double> all(world_);
    comm_->allgather(I.h_scal, 8, all.data());
    double dang = 0;
    for (double x : all) dang += x;
    I.h_scal[1] = dang;
    HIP_CHECK(hipMemcpyAsync(d_dangling.data(), I.h_scal + 1, 8,
                             hipMemcpyHostToDevice, s));
  };

  // Single-GPU pull path: the fused iteration folds apply + the NEXT
  // iteration's contrib/dangling into one rank pass (bootstrap computes
  // the first dangling/contrib outside the loop); the dangling scalar
  // ping-pongs through a fixed [cur,next] pair so the sequence is
  // hipGraph-capturable with stable pointers.
  const bool fused = pull && !multi;
  auto fused_bootstrap = [&]() {
    d_dangling.zero(s);
    pr_dangling_kernel<<<grid_for(g.owned_real), kBlock, 0, s>>>(
        rank_arr.data(), g.oe_off.data(), g.owned_real, g.v_begin,
        d_dangling.data());
    pr_contrib_kernel<<<grid_for(g.owned_real), kBlock, 0, s>>>(
        rank_arr.data(), g.oe_off.data(), g.owned_real, g.v_begin,
        contrib.data() + g.v_begin);
  };
  auto fused_iteration = [&]() {
    if (tiled) {
      acc.zero(s);
      static int tile_grid_f = [] {
        const char* e = getenv("GRAPEHIP_PR_GRID");
        return e ? atoi(e) : 8192;
      }();
      if (g.pr_total)
        pr_tiled_pull_kernel<<<tile_grid_f, kBlock, 0, s>>>(
            g.pr_tiles.data(), 0, g.pr_total, contrib.data(), acc.data());
    } else {
      pulls(0);
    }
    HIP_CHECK(hipMemsetAsync(d_dangling.data() + 1, 0, 8, s));
    if (tol > 0) d_l1.zero(s);
    pr_fused_apply_kernel<<<grid_for(owned), kBlock, 0, s>>>(
        rank_arr.data(), acc.data(), d_dangling.data(), inv_n, damping,
        g.v_begin, owned, g.owned_real, g.oe_off.data(),
        contrib.data() + g.v_begin, d_dangling.data() + 1,
        tol > 0 ? d_l1.data() : nullptr);
    HIP_CHECK(hipMemcpyAsync(d_dangling.data(), d_dangling.data() + 1, 8,
                             hipMemcpyDeviceToDevice, s));
  };

  // one iteration, recorded as a stream of kernels. Single-GPU fixed-iter
  // runs capture it into a hipGraph once and replay (the iteration is
  // fully device-side: the dangling sum feeds pr_apply through a device
  // scalar). Multi-GPU pull runs overlap compute with communication: the
  // owned-span pull (phase 1) executes on the compute stream WHILE the
  // mirror exchange of remote contribs rides the comm stream over xGMI
  // and the host folds the dangling scalar — then phase 2 accumulates
  // the remote spans (north-star overlap; ROADMAP r01 design, landed).
  auto record_iteration = [&]() {
    d_dangling.zero(s);
    pr_dangling_kernel<<<grid_for(g.owned_real), kBlock, 0, s>>>(
        rank_arr.data(), g.oe_off.data(), g.owned_real, g.v_begin,
        d_dangling.data());
    if (multi) {
      HIP_CHECK(hipMemcpyAsync(I.h_scal, d_dangling.data(), 8,
                               hipMemcpyDeviceToHost, s));
      I.ev_scal.record(s);
    }
    pr_contrib_kernel<<<grid_for(g.owned_real), kBlock, 0, s>>>(
        rank_arr.data(), g.oe_off.data(), g.owned_real, g.v_begin,
        contrib.data() + g.v_begin);
    if (pull && tiled) {
      // tiled pull: edge stream grouped by dst tile (L2-hot gathers);
      // accumulates, so clear acc first. Phase A sweeps the tiles whose
      // dsts lie in the owned slice while the mirror exchange runs.
      acc.zero(s);
      static int tile_grid = [] {
        const char* e = getenv("GRAPEHIP_PR_GRID");
        return e ? atoi(e) : 8192;
      }();
      if (multi) {
        mirror_sync_begin(I, g, contrib.data(), s);
        if (g.pr_own_hi > g.pr_own_lo)
          pr_tiled_pull_kernel<<<tile_grid, kBlock, 0, s>>>(
              g.pr_tiles.data(), g.pr_own_lo, g.pr_own_hi, contrib.data(),
              acc.data());
        fold_dangling();
        mirror_sync_end(I, g, contrib.data(), s);
        if (g.pr_own_lo > 0)
          pr_tiled_pull_kernel<<<tile_grid, kBlock, 0, s>>>(
              g.pr_tiles.data(), 0, g.pr_own_lo, contrib.data(),
              acc.data());
        if (g.pr_total > g.pr_own_hi)
          pr_tiled_pull_kernel<<<tile_grid, kBlock, 0, s>>>(
              g.pr_tiles.data(), g.pr_own_hi, g.pr_total, contrib.data(),
              acc.data());
      } else {
        if (g.pr_total)
          pr_tiled_pull_kernel<<<tile_grid, kBlock, 0, s>>>(
              g.pr_tiles.data(), 0, g.pr_total, contrib.data(),
              acc.data());
      }
    } else if (pull) {
      if (multi) {
        mirror_sync_begin(I, g, contrib.data(), s);
        pulls(1);          // owned spans, overlapped with the exchange
        fold_dangling();   // host work, overlapped too
        mirror_sync_end(I, g, contrib.data(), s);
        pulls(2);          // remote spans once payloads landed
      } else {
        pulls(0);
      }
    } else {
      if (multi) fold_dangling();
      acc.zero(s);
      int nchunks = static_cast<int>((owned + kBlock - 1) / kBlock);
      expand_cm_range<false, PrPushOp>
          <<<std::min(nchunks, kMaxGrid), kBlock, 0, s>>>(
              view, PrPushOp{contrib.data() + g.v_begin, acc.data(),
                             g.v_begin});
      if (multi) I.dc.reduce_scatter_sum_f64(acc.data(), slice, s);
    }
    if (tol > 0) d_l1.zero(s);
    pr_apply_kernel<<<grid_for(owned), kBlock, 0, s>>>(
        rank_arr.data(), acc.data(), d_dangling.data(), inv_n, damping,
        g.v_begin, owned, tol > 0 ? d_l1.data() : nullptr);
  };

  // capture once per graph (cached on the DeviceGraph: the instantiate
  // cost would otherwise swamp small runs; warmup absorbs it)
  const bool capture = !multi && tol <= 0 && g.pr_calls++ > 0;
  hipGraphExec_t graph_exec =
      reinterpret_cast<hipGraphExec_t>(g.pr_graph_exec);
  if (fused) fused_bootstrap();
  if (capture && (!graph_exec || g.pr_graph_damping != damping)) {
    if (graph_exec) {
      HIP_CHECK(hipGraphExecDestroy(graph_exec));
      graph_exec = nullptr;
    }
    hipGraph_t graph = nullptr;
    HIP_CHECK(hipStreamBeginCapture(s, hipStreamCaptureModeThreadLocal));
    if (fused) fused_iteration();
    else record_iteration();
    HIP_CHECK(hipStreamEndCapture(s, &graph));
    HIP_CHECK(hipGraphInstantiate(&graph_exec, graph, nullptr, nullptr, 0));
    HIP_CHECK(hipGraphDestroy(graph));
    g.pr_graph_exec = graph_exec;
    g.pr_graph_damping = damping;
  }

  for (int it = 0; it < iters; ++it) {
    if (capture)
      HIP_CHECK(hipGraphLaunch(graph_exec, s));
    else if (fused)
      fused_iteration();
    else
      record_iteration();
    ++rounds;
    if (tol > 0) {
      // competitor-equivalent convergence (reference pagerank_local.h):
      // stop when the global L1 delta drops under tol
      double l1 = 0;
      HIP_CHECK(hipMemcpyAsync(&l1, d_l1.data(), 8, hipMemcpyDeviceToHost,
                               s));
      HIP_CHECK(hipStreamSynchronize(s));
      if (multi) {
        std::vector<double> all(world_);
        comm_->allgather(&l1, 8, all.data());
        l1 = 0;
        for (double x : all) l1 += x;
      }
      if (l1 < tol) break;
    }
  }
  HIP_CHECK(hipDeviceSynchronize());
  if (comm_) comm_->barrier();
  double t1 = wall_s();

  GpuRunResult res;
  res.bytes_p2p = I.dc.bytes_p2p - b_p2p0;
  res.bytes_coll = I.dc.bytes_coll - b_coll0;
  res.rounds = rounds;
  res.seconds = comm_ ? comm_->allreduce_max_double(t1 - t0) : (t1 - t0);
  res.traversed_edges = static_cast<uint64_t>(iters) * g.total_edges;
  if (fetch) {
    res.f64.resize(owned);
    HIP_CHECK(hipMemcpyAsync(res.f64.data(), rank_arr.data() + g.v_begin,
                             owned * 8, hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
  }
  return res;
}

// ---------------------------------------------------------------------------
// WCC (min-root union-find, replicated parent + allreduce-min merge)
// ---------------------------------------------------------------------------
GpuRunResult GpuContext::wcc(DeviceGraph& g, bool fetch) {
  RangeMarker _mk("grapehip::wcc");
  auto& I = *impl_;
  hipStream_t s = I.compute;
  DevGraphView view = make_view(g, rank_, world_);
  uint32_t nv_pad = padded_nv(g, world_);
  uint32_t owned = g.owned();
  bool multi = world_ > 1;
  // Directed graphs: the Afforest giant-skip is only sound when every
  // vertex can see all edges that touch it. An edge u->v stored solely in
  // a skipped giant row u may be the ONLY link to v — with an in-CSR the
  // rest pass co-expands v's in-edges to cover it; without one the skip
  // is disabled (full fixpoint passes).
  const bool rest_use_in = g.directed && g.has_in;
  DevGraphView view_in = view;
  if (rest_use_in) {
    view_in.oe_off = g.ie_off.data();
    view_in.oe_dst = g.ie_dst.data();
    view_in.oe_w = nullptr;
  }

  DeviceBuffer<uint32_t> parent(nv_pad);
  DeviceBuffer<int> d_changed(1);
  // multi-GPU sparse-merge scratch: entries hooked this round
  size_t chg_words = (static_cast<size_t>(nv_pad) + 31) / 32;
  DeviceBuffer<uint32_t> chg_bm(multi ? chg_words : 1);
  DeviceBuffer<uint32_t> chg_q;
  DeviceBuffer<uint32_t> pairs_all;

  if (comm_) comm_->barrier();
  HIP_CHECK(hipDeviceSynchronize());
  double t0 = wall_s();
  const uint64_t b_p2p0 = I.dc.bytes_p2p, b_coll0 = I.dc.bytes_coll;

  iota_kernel<<<grid_for(nv_pad), kBlock, 0, s>>>(parent.data(), 0, nv_pad);
  if (multi) chg_bm.zero(s);
  int rounds = 0;
  constexpr int kSample = 2;
  uint32_t* marks = multi ? chg_bm.data() : nullptr;
  // scratch for the remaining-rows frontier
  DeviceBuffer<uint32_t> rest_bm((owned + 31) / 32 + 1);
  DeviceBuffer<uint32_t> rest_q(owned ? owned : 1);
  rest_bm.zero(s);
  for (;;) {
    int local_changed_any = 0;
    // 1) Afforest sampling: hook edge i per row in pass i, compressing
    //    between passes (linking via compressed roots collapses the giant
    //    component far better than hooking both edges in one pass)
    d_changed.zero(s);
    for (int which = 0; which < kSample; ++which) {
      if (owned)
        wcc_sample_kernel<<<grid_for(owned), kBlock, 0, s>>>(
            view, which, parent.data(), d_changed.data(), marks);
      wcc_compress_kernel<<<grid_for(nv_pad), kBlock, 0, s>>>(parent.data(),
                                                              nv_pad);
    }
    ++rounds;
    if (multi) {
      int ch0 = 0;
      HIP_CHECK(hipMemcpyAsync(&ch0, d_changed.data(), 4,
                               hipMemcpyDeviceToHost, s));
      HIP_CHECK(hipStreamSynchronize(s));
      if (ch0) local_changed_any = 1;
    }
    // 2) identify the giant component from a small host probe
    uint32_t giant = 0;
    {
      const int kProbe = 509;
      std::vector<uint32_t> roots(kProbe);
      uint32_t step = nv_pad / kProbe ? nv_pad / kProbe : 1;
      for (int i = 0; i < kProbe; ++i)
        HIP_CHECK(hipMemcpyAsync(
            &roots[i],
            parent.data() + (static_cast<uint64_t>(i) * step) % nv_pad, 4,
            hipMemcpyDeviceToHost, s));
      HIP_CHECK(hipStreamSynchronize(s));
      std::unordered_map<uint32_t, int> freq;
      int best = 0;
      for (uint32_t r : roots) {
        int c = ++freq[r];
        if (c > best) {
          best = c;
          giant = r;
        }
      }
    }
    if (g.directed && !g.has_in) giant = 0xFFFFFFFFu;  // skip unsound
    // 3) full pass restricted to rows outside the giant component,
    //    repeated to the local fixpoint
    for (;;) {
      rest_bm.zero(s);
      if (owned) {
        wcc_mark_rest_kernel<<<grid_for(owned), kBlock, 0, s>>>(
            parent.data(), view.oe_off, owned, g.v_begin, giant,
            DevBitmap{rest_bm.data()});
        if (rest_use_in)
          wcc_mark_rest_kernel<<<grid_for(owned), kBlock, 0, s>>>(
              parent.data(), view_in.oe_off, owned, g.v_begin, giant,
              DevBitmap{rest_bm.data()});
      }
      uint64_t nrest = compact_frontier(I, rest_bm.data(), owned, g.v_begin,
                                        rest_q.data(), s);
      if (nrest == 0) break;
      d_changed.zero(s);
      expand_frontier(I, view, rest_q.data(), static_cast<uint32_t>(nrest),
                      WccOp{parent.data(), d_changed.data(), marks}, s);
      if (rest_use_in)
        expand_frontier(I, view_in, rest_q.data(),
                        static_cast<uint32_t>(nrest),
                        WccOp{parent.data(), d_changed.data(), marks}, s);
      int ch = 0;
      HIP_CHECK(hipMemcpyAsync(&ch, d_changed.data(), 4,
                               hipMemcpyDeviceToHost, s));
      HIP_CHECK(hipStreamSynchronize(s));
      wcc_compress_kernel<<<grid_for(nv_pad), kBlock, 0, s>>>(parent.data(),
                                                              nv_pad);
      ++rounds;
      if (!ch) break;
      local_changed_any = 1;
    }
    if (!multi) break;
    bool any = comm_->allreduce_or(local_changed_any != 0);
    // sparse merge: ship only (entry, parent) pairs hooked this round,
    // broadcast to every peer (hooks target arbitrary global entries so
    // all ranks need them); the full allreduce-min is kept as the dense
    // fallback for churn-heavy early rounds. Compression shortcuts are
    // derived state and need no exchange — chains stay valid, and the
    // final merge below restores exact agreement (VERDICT r01 item 2:
    // per-round volume now scales with hook count, not V*world).
    {
      if (I.frontier_deg.size() < chg_words)
        I.frontier_deg.resize(chg_words + (chg_words >> 2) + 64);
      if (I.frontier_off.size() < chg_words + 1)
        I.frontier_off.resize(chg_words + (chg_words >> 2) + 65);
      popc_words_kernel<<<grid_for(chg_words), kBlock, 0, s>>>(
          chg_bm.data(), chg_words, I.frontier_deg.data());
      uint64_t nchg = exclusive_scan(I.frontier_deg.data(),
                                     I.frontier_off.data(), chg_words, s,
                                     I.scan);
      std::vector<uint64_t> counts(world_);
      comm_->allgather(&nchg, 8, counts.data());
      uint64_t total_pairs = 0;
      for (uint64_t c : counts) total_pairs += c;
      if (total_pairs * 2 >= nv_pad) {
        chg_bm.zero(s);
        I.dc.allreduce_min_u32(parent.data(), nv_pad, s);
      } else if (total_pairs) {
        std::vector<uint64_t> region(world_ + 1, 0);
        for (int f = 0; f < world_; ++f)
          region[f + 1] = region[f] + counts[f];
        if (chg_q.size() < nchg) chg_q.resize(nchg + 64);
        if (pairs_all.size() < total_pairs * 2)
          pairs_all.resize(total_pairs * 2 + 64);
        if (nchg) {
          fill_frontier_kernel<<<grid_for(chg_words), kBlock, 0, s>>>(
              chg_bm.data(), chg_words, I.frontier_off.data(), 0,
              chg_q.data());
          wcc_pack_pairs_kernel<<<grid_for(nchg), kBlock, 0, s>>>(
              chg_q.data(), nchg, parent.data(),
              pairs_all.data() + region[rank_] * 2);
        }
        std::vector<uint64_t> region_u32(world_ + 1);
        for (int f = 0; f <= world_; ++f) region_u32[f] = region[f] * 2;
        I.dc.bcast_regions_u32(pairs_all.data(), region_u32, s);
        wcc_apply_pairs_kernel<<<grid_for(total_pairs), kBlock, 0, s>>>(
            pairs_all.data(), total_pairs, region[rank_],
            region[rank_ + 1], parent.data());
      }
    }
    wcc_compress_kernel<<<grid_for(nv_pad), kBlock, 0, s>>>(parent.data(),
                                                            nv_pad);
    HIP_CHECK(hipStreamSynchronize(s));
    if (!any) break;
  }
  // exact labels for output. Multi: one dense merge first so ranks agree.
  // All paths: compress to the FIXPOINT — the loop can exit via
  // nrest==0 with entries still multiple hops from their root (path
  // halving races decide which), and the fetch below reads parent raw.
  if (multi) I.dc.allreduce_min_u32(parent.data(), nv_pad, s);
  {
    int passes = 0;
    for (;;) {
      d_changed.zero(s);
      wcc_finalize_kernel<<<grid_for(nv_pad), kBlock, 0, s>>>(
          parent.data(), nv_pad, d_changed.data());
      ++passes;
      int ch = 0;
      HIP_CHECK(hipMemcpyAsync(&ch, d_changed.data(), 4,
                               hipMemcpyDeviceToHost, s));
      HIP_CHECK(hipStreamSynchronize(s));
      if (!ch) break;
    }
    if (getenv("GRAPEHIP_DEBUG"))
      fprintf(stderr, "[wcc] output fixpoint passes=%d rounds=%d\n", passes,
              rounds);
  }
  HIP_CHECK(hipDeviceSynchronize());
  if (comm_) comm_->barrier();
  double t1 = wall_s();

  GpuRunResult res;
  res.bytes_p2p = I.dc.bytes_p2p - b_p2p0;
  res.bytes_coll = I.dc.bytes_coll - b_coll0;
  res.rounds = rounds;
  res.seconds = comm_ ? comm_->allreduce_max_double(t1 - t0) : (t1 - t0);
  res.traversed_edges = g.input_edges;
  if (fetch) {
    std::vector<uint32_t> lab(owned);
    const uint32_t* lab_src = parent.data() + g.v_begin;
    DeviceBuffer<uint32_t> relab;
    if (g.permuted) {
      // canonical labels = component min OLD id (oracle/LDBC convention)
      DeviceBuffer<uint32_t> mo(nv_pad);
      mo.fill_bytes(0xFF, s);
      wcc_minold_kernel<<<grid_for(nv_pad), kBlock, 0, s>>>(
          parent.data(), g.inv.data(), nv_pad, mo.data());
      relab.resize(owned);
      wcc_relabel_kernel<<<grid_for(owned), kBlock, 0, s>>>(
          parent.data(), mo.data(), g.v_begin, owned, relab.data());
      lab_src = relab.data();
    }
    HIP_CHECK(hipMemcpyAsync(lab.data(), lab_src, owned * 4,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    res.i64.assign(lab.begin(), lab.end());
  }
  return res;
}

// ---------------------------------------------------------------------------
// CDLP (synchronous label propagation; reference examples/analytical_apps/
// cuda/cdlp/cdlp.h). MI355X design: instead of the reference's CUB segmented
// radix sort per adjacency (cuda/cdlp/cdlp.h:199-251), the label mode is
// computed with degree-tiered counting:
//   * deg <= 64:      wave-per-row ballot mode (no memory traffic at all —
//                     the wave votes labels off with __ballot/popcount)
//   * deg <= 4096:    block-per-row open-addressing hash in LDS (64 KB:
//                     8192 slots x (label,count)), LDS atomics
//   * heavier rows:   block-per-row global-memory hash (L2 atomics), table
//                     space carved from one scratch pool by prefix sum
// Labels are u32 global vids (identity oids); each iteration refreshes the
// replicated label array with an in-place ncclAllGather over xGMI (the
// dense analogue of the reference's SendMsgThroughOEdges label push).
// Directed graphs use the in+out multiset (both CSRs), like the CPU app.
// ---------------------------------------------------------------------------

constexpr uint32_t kCdlpTinyDeg = 16;      // thread-per-row register tier
constexpr uint32_t kCdlpSmallDeg = 64;     // wave-ballot tier bound
constexpr uint32_t kCdlpWaveSlots = 1024;  // wave-hash tier (4 rows/block)
constexpr uint32_t kCdlpWaveDeg = kCdlpWaveSlots / 2;
constexpr uint32_t kCdlpLdsSlots = 8192;   // 64 KB LDS table (mid tier)
constexpr uint32_t kCdlpMidDeg = kCdlpLdsSlots / 2;  // load factor <= 0.5
constexpr uint32_t kCdlpEmpty = 0xFFFFFFFFu;

__device__ __forceinline__ uint32_t cdlp_hash(uint32_t x) {
  x ^= x >> 16;
  x *= 0x7feb352du;
  x ^= x >> 15;
  x *= 0x846ca68bu;
  x ^= x >> 16;
  return x;
}

// combined-degree bucketing over (out [+ in]) adjacency.
// tiny rows get thread-per-row register-mode treatment (tn nullptr folds
// them into the small/wave tier — the LCC reuse keeps three tiers).
__global__ void cdlp_bucket_kernel(const uint64_t* __restrict__ off1,
                                   const uint64_t* __restrict__ off2,
                                   uint32_t owned, uint32_t* tn,
                                   unsigned long long* ct, uint32_t* sm,
                                   unsigned long long* cs, uint32_t* wv,
                                   unsigned long long* cw, uint32_t* md,
                                   unsigned long long* cm, uint32_t* lg,
                                   unsigned long long* cl) {
  __shared__ uint32_t s_cnt[5];
  __shared__ unsigned long long s_base[5];
  uint32_t* lists[5] = {tn, sm, wv, md, lg};
  unsigned long long* gcnt[5] = {ct, cs, cw, cm, cl};
  const uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t base = blockIdx.x * blockDim.x; base < owned;
       base += stride) {
    if (threadIdx.x < 5) s_cnt[threadIdx.x] = 0;
    __syncthreads();
    uint32_t r = base + threadIdx.x;
    int b = -1;
    uint32_t loc = 0;
    if (r < owned) {
      uint64_t deg = off1[r + 1] - off1[r];
      if (off2) deg += off2[r + 1] - off2[r];
      if (tn && deg <= kCdlpTinyDeg)
        b = 0;
      else if (deg <= kCdlpSmallDeg)
        b = 1;
      else if (wv && deg <= kCdlpWaveDeg)
        b = 2;
      else if (deg <= kCdlpMidDeg)
        b = 3;
      else
        b = 4;
      loc = atomicAdd(&s_cnt[b], 1u);
    }
    __syncthreads();
    if (threadIdx.x < 5 && s_cnt[threadIdx.x])
      s_base[threadIdx.x] =
          atomicAdd(gcnt[threadIdx.x],
                    static_cast<unsigned long long>(s_cnt[threadIdx.x]));
    __syncthreads();
    if (b >= 0) lists[b][s_base[b] + loc] = r;
    __syncthreads();
  }
}

// thread-per-row register mode for deg <= 8 (most power-law rows):
// a wave-per-row ballot on a deg-3 row idles 61 lanes
__global__ void cdlp_tiny_kernel(const uint64_t* __restrict__ off1,
                                 const uint32_t* __restrict__ dst1,
                                 const uint64_t* __restrict__ off2,
                                 const uint32_t* __restrict__ dst2,
                                 const uint32_t* __restrict__ lab,
                                 const uint32_t* __restrict__ rows,
                                 uint64_t nrows, uint32_t v_begin,
                                 const uint32_t* __restrict__ dirty,
                                 uint32_t* __restrict__ next) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < nrows; i += stride) {
    uint32_t r = rows[i];
    if (dirty) {
      uint32_t vg = v_begin + r;
      if (!((dirty[vg >> 5] >> (vg & 31)) & 1)) continue;
    }
    uint64_t b1 = off1[r], e1 = off1[r + 1];
    uint32_t d1 = static_cast<uint32_t>(e1 - b1);
    uint32_t deg = d1;
    uint64_t b2 = 0;
    if (off2) {
      b2 = off2[r];
      deg += static_cast<uint32_t>(off2[r + 1] - b2);
    }
    if (deg == 0) {
      next[r] = lab[v_begin + r];
      continue;
    }
    uint32_t l[kCdlpTinyDeg];
#pragma unroll
    for (uint32_t k = 0; k < kCdlpTinyDeg; ++k)
      if (k < deg)
        l[k] = k < d1 ? lab[dst1[b1 + k]] : lab[dst2[b2 + (k - d1)]];
    uint32_t best_lab = kCdlpEmpty, best_cnt = 0;
    for (uint32_t k = 0; k < deg; ++k) {
      bool first = true;
      for (uint32_t j = 0; j < k; ++j)
        if (l[j] == l[k]) first = false;
      if (!first) continue;
      uint32_t cnt = 1;
      for (uint32_t j = k + 1; j < deg; ++j)
        if (l[j] == l[k]) ++cnt;
      if (cnt > best_cnt || (cnt == best_cnt && l[k] < best_lab)) {
        best_cnt = cnt;
        best_lab = l[k];
      }
    }
    next[r] = best_lab;
  }
}

// wave-per-row ballot mode, deg <= 64
__global__ void cdlp_small_kernel(const uint64_t* __restrict__ off1,
                                  const uint32_t* __restrict__ dst1,
                                  const uint64_t* __restrict__ off2,
                                  const uint32_t* __restrict__ dst2,
                                  const uint32_t* __restrict__ lab,
                                  const uint32_t* __restrict__ rows,
                                  uint64_t nrows, uint32_t v_begin,
                                  const uint32_t* __restrict__ dirty,
                                  uint32_t* __restrict__ next) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wpb = kBlock / kWave;
  size_t wstride = static_cast<size_t>(gridDim.x) * wpb;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * wpb + wid; i < nrows;
       i += wstride) {
    uint32_t r = rows[i];
    if (dirty) {
      uint32_t v = v_begin + r;
      if (!((dirty[v >> 5] >> (v & 31)) & 1)) continue;
    }
    uint64_t b1 = off1[r], e1 = off1[r + 1];
    uint32_t d1 = static_cast<uint32_t>(e1 - b1);
    uint32_t deg = d1;
    uint64_t b2 = 0;
    if (off2) {
      b2 = off2[r];
      deg += static_cast<uint32_t>(off2[r + 1] - b2);
    }
    if (deg == 0) {
      if (lane == 0) next[r] = lab[v_begin + r];
      continue;
    }
    uint32_t mylab = kCdlpEmpty;
    if (lane < static_cast<int>(deg))
      mylab = lane < static_cast<int>(d1)
                  ? lab[dst1[b1 + lane]]
                  : lab[dst2[b2 + (lane - d1)]];
    unsigned long long remaining = __ballot(lane < static_cast<int>(deg));
    uint32_t best_cnt = 0, best_lab = kCdlpEmpty;
    while (remaining) {
      int L = __ffsll(static_cast<unsigned long long>(remaining)) - 1;
      uint32_t lab0 = __shfl(mylab, L, 64);
      unsigned long long m = __ballot(mylab == lab0) & remaining;
      remaining &= ~m;
      uint32_t cnt = __popcll(m);
      if (cnt > best_cnt || (cnt == best_cnt && lab0 < best_lab)) {
        best_cnt = cnt;
        best_lab = lab0;
      }
    }
    if (lane == 0) next[r] = best_lab;
  }
}

// shared argmax reduce over (count, ~label) keys
__device__ __forceinline__ void cdlp_block_argmax(uint64_t key,
                                                  uint32_t* out,
                                                  uint32_t fallback) {
  __shared__ uint64_t s_wave[kBlock / kWave];
#pragma unroll
  for (int d = 32; d > 0; d >>= 1) {
    uint64_t o = __shfl_down(static_cast<unsigned long long>(key), d, 64);
    if (o > key) key = o;
  }
  if ((threadIdx.x & 63) == 0) s_wave[threadIdx.x >> 6] = key;
  __syncthreads();
  if (threadIdx.x == 0) {
    uint64_t best = 0;
#pragma unroll
    for (int w = 0; w < kBlock / kWave; ++w)
      if (s_wave[w] > best) best = s_wave[w];
    *out = best ? ~static_cast<uint32_t>(best) : fallback;
  }
  __syncthreads();
}

// 4-rows-per-block wave hash for 64 < deg <= kCdlpWaveDeg: a block-per-row
// table on a deg-100 row leaves 3 of 4 waves idle; here each wave owns a
// 1024-slot LDS region and the group loop is uniform so plain
// __syncthreads stages the clear/insert/reduce phases.
__global__ void cdlp_wave_kernel(const uint64_t* __restrict__ off1,
                                 const uint32_t* __restrict__ dst1,
                                 const uint64_t* __restrict__ off2,
                                 const uint32_t* __restrict__ dst2,
                                 const uint32_t* __restrict__ lab,
                                 const uint32_t* __restrict__ rows,
                                 uint64_t nrows, uint32_t v_begin,
                                 const uint32_t* __restrict__ dirty,
                                 uint32_t* __restrict__ next) {
  __shared__ uint32_t s_lab[4][kCdlpWaveSlots];
  __shared__ uint32_t s_cnt[4][kCdlpWaveSlots];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  uint64_t groups = (nrows + 3) / 4;
  for (uint64_t grp = blockIdx.x; grp < groups; grp += gridDim.x) {
    uint64_t i = grp * 4 + wid;
    bool have = i < nrows;
    uint32_t r = have ? rows[i] : 0;
    if (have && dirty) {
      uint32_t vg = v_begin + r;
      if (!((dirty[vg >> 5] >> (vg & 31)) & 1)) have = false;
    }
    uint32_t d1 = 0, deg = 0;
    uint64_t b1 = 0, b2 = 0;
    uint32_t cap = 64, mask = 63;
    if (have) {
      b1 = off1[r];
      d1 = static_cast<uint32_t>(off1[r + 1] - b1);
      deg = d1;
      if (off2) {
        b2 = off2[r];
        deg += static_cast<uint32_t>(off2[r + 1] - b2);
      }
      while (cap < 2 * deg) cap <<= 1;
      if (cap > kCdlpWaveSlots) cap = kCdlpWaveSlots;
      mask = cap - 1;
      for (uint32_t k = lane; k < cap; k += kWave) {
        s_lab[wid][k] = kCdlpEmpty;
        s_cnt[wid][k] = 0;
      }
    }
    __syncthreads();
    if (have) {
      for (uint32_t k = lane; k < deg; k += kWave) {
        uint32_t l = k < d1 ? lab[dst1[b1 + k]] : lab[dst2[b2 + (k - d1)]];
        uint32_t idx = cdlp_hash(l) & mask;
        for (;;) {
          uint32_t old = atomicCAS(&s_lab[wid][idx], kCdlpEmpty, l);
          if (old == kCdlpEmpty || old == l) {
            atomicAdd(&s_cnt[wid][idx], 1u);
            break;
          }
          idx = (idx + 1) & mask;
        }
      }
    }
    __syncthreads();
    if (have) {
      uint64_t key = 0;
      for (uint32_t k = lane; k < cap; k += kWave) {
        uint32_t c = s_cnt[wid][k];
        if (c) {
          uint64_t cand = (static_cast<uint64_t>(c) << 32) |
                          static_cast<uint32_t>(~s_lab[wid][k]);
          if (cand > key) key = cand;
        }
      }
#pragma unroll
      for (int d = 32; d > 0; d >>= 1) {
        uint64_t o = __shfl_down(static_cast<unsigned long long>(key), d,
                                 64);
        if (o > key) key = o;
      }
      if (lane == 0)
        next[r] = key ? ~static_cast<uint32_t>(key)
                      : (deg ? 0 : lab[v_begin + r]);
    }
    __syncthreads();
  }
}

// block-per-row LDS hash, deg <= kCdlpMidDeg
__global__ void cdlp_mid_kernel(const uint64_t* __restrict__ off1,
                                const uint32_t* __restrict__ dst1,
                                const uint64_t* __restrict__ off2,
                                const uint32_t* __restrict__ dst2,
                                const uint32_t* __restrict__ lab,
                                const uint32_t* __restrict__ rows,
                                uint64_t nrows, uint32_t v_begin,
                                const uint32_t* __restrict__ dirty,
                                uint32_t* __restrict__ next) {
  __shared__ uint32_t s_lab[kCdlpLdsSlots];
  __shared__ uint32_t s_cnt[kCdlpLdsSlots];
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t r = rows[i];
    if (dirty) {
      uint32_t vg = v_begin + r;
      if (!((dirty[vg >> 5] >> (vg & 31)) & 1)) continue;
    }
    uint64_t b1 = off1[r], e1 = off1[r + 1];
    uint32_t d1 = static_cast<uint32_t>(e1 - b1);
    uint32_t deg = d1;
    uint64_t b2 = 0;
    if (off2) {
      b2 = off2[r];
      deg += static_cast<uint32_t>(off2[r + 1] - b2);
    }
    // capacity: pow2 >= 2*deg, <= kCdlpLdsSlots
    uint32_t cap = 64;
    while (cap < 2 * deg) cap <<= 1;
    if (cap > kCdlpLdsSlots) cap = kCdlpLdsSlots;
    const uint32_t mask = cap - 1;
    for (uint32_t k = threadIdx.x; k < cap; k += blockDim.x) {
      s_lab[k] = kCdlpEmpty;
      s_cnt[k] = 0;
    }
    __syncthreads();
    for (uint32_t k = threadIdx.x; k < deg; k += blockDim.x) {
      uint32_t l = k < d1 ? lab[dst1[b1 + k]] : lab[dst2[b2 + (k - d1)]];
      uint32_t idx = cdlp_hash(l) & mask;
      for (;;) {
        uint32_t old = atomicCAS(&s_lab[idx], kCdlpEmpty, l);
        if (old == kCdlpEmpty || old == l) {
          atomicAdd(&s_cnt[idx], 1u);
          break;
        }
        idx = (idx + 1) & mask;
      }
    }
    __syncthreads();
    uint64_t key = 0;
    for (uint32_t k = threadIdx.x; k < cap; k += blockDim.x) {
      uint32_t c = s_cnt[k];
      if (c) {
        uint64_t cand = (static_cast<uint64_t>(c) << 32) |
                        static_cast<uint32_t>(~s_lab[k]);
        if (cand > key) key = cand;
      }
    }
    cdlp_block_argmax(key, &next[r], lab[v_begin + r]);
  }
}

// per-heavy-row table capacity (pow2 >= 2*deg)
__global__ void cdlp_heavy_cap_kernel(const uint64_t* __restrict__ off1,
                                      const uint64_t* __restrict__ off2,
                                      const uint32_t* __restrict__ rows,
                                      uint64_t nrows,
                                      uint32_t* __restrict__ caps) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < nrows; i += stride) {
    uint32_t r = rows[i];
    uint64_t deg = off1[r + 1] - off1[r];
    if (off2) deg += off2[r + 1] - off2[r];
    uint64_t cap = 1024;
    while (cap < 2 * deg) cap <<= 1;
    caps[i] = static_cast<uint32_t>(cap);
  }
}

// block-per-row global hash (heavy rows). Slots are epoch-tagged: the
// count word packs (epoch << 24 | count), so tables never need clearing
// between iterations — a stale slot (old epoch) reads as empty.
__global__ void cdlp_large_kernel(const uint64_t* __restrict__ off1,
                                  const uint32_t* __restrict__ dst1,
                                  const uint64_t* __restrict__ off2,
                                  const uint32_t* __restrict__ dst2,
                                  const uint32_t* __restrict__ lab,
                                  const uint32_t* __restrict__ rows,
                                  uint64_t nrows,
                                  const uint64_t* __restrict__ tbl_off,
                                  unsigned long long* __restrict__ tbl,
                                  uint32_t epoch, uint32_t v_begin,
                                  const uint32_t* __restrict__ dirty,
                                  uint32_t* __restrict__ next) {
  // slot u64 = (epoch:16 | label:32 hashed into low? ) — layout:
  // high 16 bits epoch, next 32 bits label, low 16 bits... counts can
  // exceed 16 bits, so: slot = (epoch<<48) | (label<<16) is unsafe.
  // Instead: two u32 halves in one u64: hi = label, lo = (epoch<<24|cnt)
  // with cnt capped at 2^24 (heavy rows cap the count at deg < 2^24 per
  // label; larger multiplicities clamp — mode selection unaffected since
  // clamp only at astronomically heavy rows).
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t r = rows[i];
    if (dirty) {
      uint32_t vg = v_begin + r;
      if (!((dirty[vg >> 5] >> (vg & 31)) & 1)) continue;
    }
    uint64_t b1 = off1[r], e1 = off1[r + 1];
    uint64_t d1 = e1 - b1;
    uint64_t deg = d1;
    uint64_t b2 = 0;
    if (off2) {
      b2 = off2[r];
      deg += off2[r + 1] - b2;
    }
    uint64_t tb = tbl_off[i];
    const uint64_t cap = tbl_off[i + 1] - tb;
    const uint64_t mask = cap - 1;
    unsigned long long* t = tbl + tb;
    const uint32_t etag = epoch << 24;
    for (uint64_t k = threadIdx.x; k < deg; k += blockDim.x) {
      uint32_t l = k < d1 ? lab[dst1[b1 + k]] : lab[dst2[b2 + (k - d1)]];
      uint64_t idx = cdlp_hash(l) & mask;
      for (;;) {
        unsigned long long cur = t[idx];
        uint32_t cur_lab = static_cast<uint32_t>(cur >> 32);
        uint32_t cur_lo = static_cast<uint32_t>(cur);
        bool stale = (cur_lo >> 24) != epoch;
        if (stale) {
          // try to claim the slot for this epoch with count 1
          unsigned long long want =
              (static_cast<unsigned long long>(l) << 32) | (etag | 1u);
          unsigned long long old = atomicCAS(&t[idx], cur, want);
          if (old == cur) break;      // claimed
          cur = old;                  // somebody else raced; re-inspect
          cur_lab = static_cast<uint32_t>(cur >> 32);
          cur_lo = static_cast<uint32_t>(cur);
          if ((cur_lo >> 24) != epoch) continue;  // still stale: retry slot
        }
        if (cur_lab == l) {
          if ((cur_lo & 0xFFFFFF) != 0xFFFFFF) atomicAdd(&t[idx], 1ull);
          break;
        }
        idx = (idx + 1) & mask;
      }
    }
    __syncthreads();
    uint64_t key = 0;
    for (uint64_t k = threadIdx.x; k < cap; k += blockDim.x) {
      unsigned long long cur = t[k];
      uint32_t lo = static_cast<uint32_t>(cur);
      if ((lo >> 24) == epoch && (lo & 0xFFFFFF)) {
        uint64_t cand =
            (static_cast<uint64_t>(lo & 0xFFFFFF) << 32) |
            static_cast<uint32_t>(~static_cast<uint32_t>(cur >> 32));
        if (cand > key) key = cand;
      }
    }
    cdlp_block_argmax(key, &next[r], lab[v_begin + r]);
  }
}

__global__ void cdlp_commit_kernel(const uint32_t* __restrict__ next,
                                   uint32_t owned, uint32_t v_begin,
                                   const uint32_t* __restrict__ dirty,
                                   uint32_t* __restrict__ lab,
                                   DevBitmap changed,
                                   unsigned long long* __restrict__ nch) {
  __shared__ unsigned long long s_nch;
  if (threadIdx.x == 0) s_nch = 0;
  __syncthreads();
  unsigned long long my = 0;
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < owned;
       r += stride) {
    uint32_t v = v_begin + r;
    if (dirty && !((dirty[v >> 5] >> (v & 31)) & 1)) continue;
    uint32_t nv = next[r];
    if (nv != lab[v]) {
      lab[v] = nv;
      changed.set_once(r);
      ++my;
    }
  }
  if (my) atomicAdd(&s_nch, my);
  __syncthreads();
  if (threadIdx.x == 0 && s_nch) atomicAdd(nch, s_nch);
}

// mark rows whose label multiset can have changed: every neighbor of a
// changed vertex (undirected: out-CSR covers it; directed: out + in)
struct CdlpMarkOp {
  DevBitmap dirty;
  __device__ __forceinline__ void operator()(uint32_t v, uint32_t) const {
    dirty.set_once(v);
  }
};

struct CdlpDirtyOp {
  DevBitmap dirty;  // global-vid bitmap
  uint32_t v_begin, v_end;
  bool multi;
  DevHalo halo;
  __device__ __forceinline__ void operator()(uint32_t, uint32_t d,
                                             float) const {
    if (d >= v_begin && d < v_end) {
      dirty.set_once(d);
    } else if (multi) {
      halo.add(d);
    } else {
      dirty.set_once(d);
    }
  }
};

GpuRunResult GpuContext::cdlp(DeviceGraph& g, int iters, bool fetch) {
  RangeMarker _mk("grapehip::cdlp");
  auto& I = *impl_;
  hipStream_t s = I.compute;
  uint32_t nv_pad = padded_nv(g, world_);
  uint32_t owned = g.owned();
  bool multi = world_ > 1;
  if (g.directed && !g.has_in)
    throw std::runtime_error(
        "GPU CDLP on a directed graph needs the in-CSR "
        "(load with build_in_csr=True)");

  const uint64_t* off1 = g.oe_off.data();
  const uint32_t* dst1 = g.oe_dst.data();
  const uint64_t* off2 = g.directed ? g.ie_off.data() : nullptr;
  const uint32_t* dst2 = g.directed ? g.ie_dst.data() : nullptr;

  // tier the rows once (degrees don't change across iterations)
  DeviceBuffer<uint32_t> t_tiny(owned), t_small(owned), t_wave(owned),
      t_mid(owned), t_large(owned);
  DeviceBuffer<unsigned long long> cnts(5);
  cnts.zero(s);
  if (owned)
    cdlp_bucket_kernel<<<grid_for(owned), kBlock, 0, s>>>(
        off1, off2, owned, t_tiny.data(), cnts.data() + 0, t_small.data(),
        cnts.data() + 1, t_wave.data(), cnts.data() + 2, t_mid.data(),
        cnts.data() + 3, t_large.data(), cnts.data() + 4);
  auto hc = cnts.download(s);
  uint64_t n_tiny = hc[0], n_small = hc[1], n_wave = hc[2], n_mid = hc[3],
           n_large = hc[4];

  // heavy-row global hash pool (epoch-tagged u64 slots: no per-iteration
  // clears; epoch 0 == the zeroed virgin state)
  DeviceBuffer<uint32_t> heavy_caps;
  DeviceBuffer<uint64_t> heavy_off;
  DeviceBuffer<unsigned long long> heavy_tbl;
  uint64_t heavy_total = 0;
  if (n_large) {
    heavy_caps.resize(n_large);
    heavy_off.resize(n_large + 1);
    cdlp_heavy_cap_kernel<<<grid_for(n_large), kBlock, 0, s>>>(
        off1, off2, t_large.data(), n_large, heavy_caps.data());
    heavy_total = exclusive_scan(heavy_caps.data(), heavy_off.data(),
                                 n_large, s, I.scan);
    heavy_tbl.resize(heavy_total);
    heavy_tbl.zero(s);
  }

  DeviceBuffer<uint32_t> lab(nv_pad);
  DeviceBuffer<uint32_t> next(owned ? owned : 1);
  // delta-convergence machinery: only rows with a changed neighborhood
  // recompute their mode (exact: an unchanged multiset reproduces the
  // same mode); fixpoint exits early
  size_t dirty_words = (static_cast<size_t>(nv_pad) + 31) / 32 + 1;
  DeviceBuffer<uint32_t> dirty(dirty_words);
  DeviceBuffer<uint32_t> changed_bm((owned + 31) / 32 + 1);
  DeviceBuffer<uint32_t> changed_q(owned ? owned : 1);
  DeviceBuffer<unsigned long long> d_nch(1);
  uint64_t halo_cap = nv_pad / (world_ ? world_ : 1);
  if (multi) {
    I.halo_idx.resize(static_cast<uint64_t>(world_) * halo_cap);
    I.halo_cnt.resize(world_);
    I.halo_bm.resize(dirty_words);
    I.halo_cnt.zero(s);
    I.halo_bm.zero(s);
  }
  DevGraphView view = make_view(g, rank_, world_);
  DevGraphView view_in = view;
  if (g.directed) {
    view_in.oe_off = g.ie_off.data();
    view_in.oe_dst = g.ie_dst.data();
    view_in.oe_w = nullptr;
  }
  if (multi) ensure_mirrors(I, comm_, g, rank_, world_, s);

  if (comm_) comm_->barrier();
  HIP_CHECK(hipDeviceSynchronize());
  double t0 = wall_s();
  const uint64_t b_p2p0 = I.dc.bytes_p2p, b_coll0 = I.dc.bytes_coll;

  // label space must compare in OID order (reference tie-break): dense
  // renumber uses the sorted-oid table, hub renumber the inverse perm
  const uint32_t* lab0 = g.oid_order.size()
                             ? g.oid_order.data()
                             : (g.permuted ? g.inv.data() : nullptr);
  if (lab0)
    HIP_CHECK(hipMemcpyAsync(lab.data(), lab0,
                             static_cast<size_t>(nv_pad) * 4,
                             hipMemcpyDeviceToDevice, s));
  else
    iota_kernel<<<grid_for(nv_pad), kBlock, 0, s>>>(lab.data(), 0, nv_pad);
  changed_bm.zero(s);
  int rounds = 0;
  bool use_dirty = false;  // iteration 0 recomputes everything
  for (int it = 0; it < iters; ++it) {
    const uint32_t* dw = use_dirty ? dirty.data() : nullptr;
    if (n_tiny)
      cdlp_tiny_kernel<<<grid_for(n_tiny), kBlock, 0, s>>>(
          off1, dst1, off2, dst2, lab.data(), t_tiny.data(), n_tiny,
          g.v_begin, dw, next.data());
    if (n_small)
      cdlp_small_kernel<<<grid_for(n_small * kWave), kBlock, 0, s>>>(
          off1, dst1, off2, dst2, lab.data(), t_small.data(), n_small,
          g.v_begin, dw, next.data());
    if (n_wave)
      cdlp_wave_kernel<<<grid_for(((n_wave + 3) / 4) * kBlock), kBlock, 0,
                         s>>>(off1, dst1, off2, dst2, lab.data(),
                              t_wave.data(), n_wave, g.v_begin, dw,
                              next.data());
    if (n_mid)
      cdlp_mid_kernel<<<std::min<int>(n_mid, kMaxGrid), kBlock, 0, s>>>(
          off1, dst1, off2, dst2, lab.data(), t_mid.data(), n_mid, g.v_begin,
          dw, next.data());
    if (n_large && it > 0 && it % 254 == 0)
      heavy_tbl.zero(s);  // epoch tag is 8-bit; re-zero on wrap
    if (n_large)
      cdlp_large_kernel<<<std::min<int>(n_large, kMaxGrid), kBlock, 0, s>>>(
          off1, dst1, off2, dst2, lab.data(), t_large.data(), n_large,
          heavy_off.data(), heavy_tbl.data(),
          static_cast<uint32_t>((it % 254) + 1), g.v_begin, dw,
          next.data());
    d_nch.zero(s);
    if (owned)
      cdlp_commit_kernel<<<grid_for(owned), kBlock, 0, s>>>(
          next.data(), owned, g.v_begin, dw, lab.data(),
          DevBitmap{changed_bm.data()}, d_nch.data());
    ++rounds;
    unsigned long long nch = 0;
    HIP_CHECK(hipMemcpyAsync(&nch, d_nch.data(), 8, hipMemcpyDeviceToHost,
                             s));
    HIP_CHECK(hipStreamSynchronize(s));
    uint64_t g_nch = multi ? comm_->allreduce_sum(nch) : nch;
    if (g_nch == 0) break;  // fixpoint: nothing changed, nothing to sync
    if (multi) {
      // refresh referenced remote labels point-to-point (per-link volume
      // scales with boundary size, not V·world); once churn decays,
      // ship only the CHANGED labels as (id, value) pairs
      uint64_t g_owned_now =
          comm_->allreduce_sum(static_cast<uint64_t>(owned));
      if (g_nch * 8 < g_owned_now) {
        mirror_sync_changed(I, comm_, g, lab.data(), changed_bm.data(),
                            rank_, world_, s);
      } else {
        mirror_sync_begin(I, g, lab.data(), s);
        mirror_sync_end(I, g, lab.data(), s);
      }
    }
    if (it + 1 == iters) break;
    // adaptive: building the dirty set costs an edge expansion over the
    // changed rows — only worth it once changes are sparse (converging
    // graphs); with heavy churn (early rounds, RMAT oscillation) a full
    // recompute is cheaper than sweep + marking
    uint64_t g_owned =
        multi ? comm_->allreduce_sum(static_cast<uint64_t>(owned))
              : owned;
    use_dirty = g_nch < g_owned / 8;
    if (!use_dirty) {
      changed_bm.zero(s);
      continue;
    }
    // rebuild the dirty set from this round's changed rows
    dirty.zero(s);
    uint64_t qn = compact_frontier(I, changed_bm.data(), owned, g.v_begin,
                                   changed_q.data(), s);
    if (qn) {
      CdlpDirtyOp op{DevBitmap{dirty.data()}, g.v_begin, g.v_end, multi,
                     DevHalo{I.halo_idx.data(), I.halo_cnt.data(),
                             DevBitmap{I.halo_bm.data()}, halo_cap,
                             view.slice, world_}};
      expand_frontier(I, view, changed_q.data(),
                      static_cast<uint32_t>(qn), op, s);
      if (g.directed)
        expand_frontier(I, view_in, changed_q.data(),
                        static_cast<uint32_t>(qn), op, s);
    }
    if (multi) {
      uint64_t nrecv = halo_flush<uint32_t>(I, comm_, rank_, world_,
                                            lab.data(), halo_cap, s);
      if (nrecv) {
        DevBitmap db{dirty.data()};
        halo_process_kernel<uint32_t, CdlpMarkOp>
            <<<grid_for(nrecv), kBlock, 0, s>>>(
                reinterpret_cast<HaloPair<uint32_t>*>(I.recvbuf.data()),
                nrecv, CdlpMarkOp{db});
      }
    }
  }
  HIP_CHECK(hipDeviceSynchronize());
  if (comm_) comm_->barrier();
  double t1 = wall_s();

  GpuRunResult res;
  res.bytes_p2p = I.dc.bytes_p2p - b_p2p0;
  res.bytes_coll = I.dc.bytes_coll - b_coll0;
  res.rounds = rounds;
  res.seconds = comm_ ? comm_->allreduce_max_double(t1 - t0) : (t1 - t0);
  res.traversed_edges =
      static_cast<uint64_t>(iters) * g.total_edges * (g.directed ? 2 : 1);
  if (fetch) {
    std::vector<uint32_t> l32(owned);
    HIP_CHECK(hipMemcpyAsync(l32.data(), lab.data() + g.v_begin, owned * 4,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    res.i64.assign(l32.begin(), l32.end());
  }
  return res;
}
// ---------------------------------------------------------------------------
// LCC (local clustering coefficient; reference examples/analytical_apps/
// cuda/lcc/lcc.h). LDBC semantics (matches the CPU app in apps/lcc.hpp):
//   D(v) = |distinct in∪out neighbors, excl. self|
//   lcc(v) = 2·tri(v) / (D(v)·(D(v)−1))
// MI355X pipeline (no CUB segmented radix — hash dedup + LDS bitonic):
//   1. D(v) per owned row via degree-tiered hash distinct-count
//      (wave-ballot / LDS hash / global hash — same tiers as CDLP)
//   2. allgather D over xGMI (replicated u32 array)
//   3. oriented adjacency O(v) = {u ∈ N(v) : (D(u),u) > (D(v),v)}, deduped
//      during the same hash pass; each edge u—v survives in exactly one
//      direction, so Σ|O| = E_simple/… and every triangle is found once
//   4. per-row ascending sort of O(v): LDS bitonic ≤4096, global scratch
//      bitonic for heavier rows (reference uses CUB SegmentedRadixSort,
//      cuda/utils/cuda_utils.h:194-290)
//   5. ragged allgather of the oriented CSR (per-rank region broadcast)
//   6. wave-per-row intersection count: for each oriented edge (u,v),
//      merge-intersect O(u)∩O(v);每 hit credits u, v, w (atomicAdd u64)
//   7. allreduce-sum T, finalize lcc = 2T/(D(D−1))
// ---------------------------------------------------------------------------

constexpr uint32_t kLccLdsSlots = 8192;  // 32 KB label-only LDS hash
constexpr uint32_t kLccSortLds = 4096;   // 16 KB LDS bitonic bound

// orientation: keep u in O(v) iff (D[u],u) > (D[v],v)
__device__ __forceinline__ bool lcc_keep(uint32_t du, uint32_t u, uint32_t dv,
                                         uint32_t v) {
  return du > dv || (du == dv && u > v);
}

// --- pass 1: distinct neighbor count -------------------------------------

__global__ void lcc_distinct_small_kernel(
    const uint64_t* __restrict__ off1, const uint32_t* __restrict__ dst1,
    const uint64_t* __restrict__ off2, const uint32_t* __restrict__ dst2,
    const uint32_t* __restrict__ rows, uint64_t nrows, uint32_t v_begin,
    uint32_t* __restrict__ D) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wpb = kBlock / kWave;
  size_t wstride = static_cast<size_t>(gridDim.x) * wpb;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * wpb + wid; i < nrows;
       i += wstride) {
    uint32_t r = rows[i];
    uint32_t v = v_begin + r;
    uint64_t b1 = off1[r], e1 = off1[r + 1];
    uint32_t d1 = static_cast<uint32_t>(e1 - b1);
    uint32_t deg = d1;
    uint64_t b2 = 0;
    if (off2) {
      b2 = off2[r];
      deg += static_cast<uint32_t>(off2[r + 1] - b2);
    }
    uint32_t u = kCdlpEmpty;
    if (lane < static_cast<int>(deg))
      u = lane < static_cast<int>(d1) ? dst1[b1 + lane]
                                      : dst2[b2 + (lane - d1)];
    unsigned long long remaining =
        __ballot(lane < static_cast<int>(deg) && u != v);
    uint32_t distinct = 0;
    while (remaining) {
      int L = __ffsll(remaining) - 1;
      uint32_t u0 = __shfl(u, L, 64);
      remaining &= ~__ballot(u == u0);
      ++distinct;
    }
    if (lane == 0) D[v] = distinct;
  }
}

__global__ void lcc_distinct_mid_kernel(
    const uint64_t* __restrict__ off1, const uint32_t* __restrict__ dst1,
    const uint64_t* __restrict__ off2, const uint32_t* __restrict__ dst2,
    const uint32_t* __restrict__ rows, uint64_t nrows, uint32_t v_begin,
    uint32_t* __restrict__ D) {
  __shared__ uint32_t s_lab[kLccLdsSlots];
  __shared__ uint32_t s_distinct;
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t r = rows[i];
    uint32_t v = v_begin + r;
    uint64_t b1 = off1[r], e1 = off1[r + 1];
    uint32_t d1 = static_cast<uint32_t>(e1 - b1);
    uint32_t deg = d1;
    uint64_t b2 = 0;
    if (off2) {
      b2 = off2[r];
      deg += static_cast<uint32_t>(off2[r + 1] - b2);
    }
    uint32_t cap = 64;
    while (cap < 2 * deg) cap <<= 1;
    if (cap > kLccLdsSlots) cap = kLccLdsSlots;
    const uint32_t mask = cap - 1;
    for (uint32_t k = threadIdx.x; k < cap; k += blockDim.x)
      s_lab[k] = kCdlpEmpty;
    if (threadIdx.x == 0) s_distinct = 0;
    __syncthreads();
    for (uint32_t k = threadIdx.x; k < deg; k += blockDim.x) {
      uint32_t u = k < d1 ? dst1[b1 + k] : dst2[b2 + (k - d1)];
      if (u == v) continue;
      uint32_t idx = cdlp_hash(u) & mask;
      for (;;) {
        uint32_t old = atomicCAS(&s_lab[idx], kCdlpEmpty, u);
        if (old == kCdlpEmpty) {
          atomicAdd(&s_distinct, 1u);
          break;
        }
        if (old == u) break;
        idx = (idx + 1) & mask;
      }
    }
    __syncthreads();
    if (threadIdx.x == 0) D[v] = s_distinct;
    __syncthreads();
  }
}

__global__ void lcc_distinct_large_kernel(
    const uint64_t* __restrict__ off1, const uint32_t* __restrict__ dst1,
    const uint64_t* __restrict__ off2, const uint32_t* __restrict__ dst2,
    const uint32_t* __restrict__ rows, uint64_t nrows,
    const uint64_t* __restrict__ tbl_off, uint32_t* __restrict__ tbl_lab,
    uint32_t v_begin, uint32_t* __restrict__ D) {
  __shared__ uint32_t s_distinct;
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t r = rows[i];
    uint32_t v = v_begin + r;
    uint64_t b1 = off1[r], e1 = off1[r + 1];
    uint64_t d1 = e1 - b1;
    uint64_t deg = d1;
    uint64_t b2 = 0;
    if (off2) {
      b2 = off2[r];
      deg += off2[r + 1] - b2;
    }
    uint64_t tb = tbl_off[i];
    const uint64_t mask = (tbl_off[i + 1] - tb) - 1;
    uint32_t* tl = tbl_lab + tb;
    if (threadIdx.x == 0) s_distinct = 0;
    __syncthreads();
    for (uint64_t k = threadIdx.x; k < deg; k += blockDim.x) {
      uint32_t u = k < d1 ? dst1[b1 + k] : dst2[b2 + (k - d1)];
      if (u == v) continue;
      uint64_t idx = cdlp_hash(u) & mask;
      for (;;) {
        uint32_t old = atomicCAS(&tl[idx], kCdlpEmpty, u);
        if (old == kCdlpEmpty) {
          atomicAdd(&s_distinct, 1u);
          break;
        }
        if (old == u) break;
        idx = (idx + 1) & mask;
      }
    }
    __syncthreads();
    if (threadIdx.x == 0) D[v] = s_distinct;
    __syncthreads();
  }
}

// 4-rows-per-block wave variants for 64 < deg <= kCdlpWaveDeg (a block-
// per-row hash on a deg-100 row idles 3 of 4 waves; same uniform-barrier
// staging as cdlp_wave_kernel)
__global__ void lcc_distinct_wave_kernel(
    const uint64_t* __restrict__ off1, const uint32_t* __restrict__ dst1,
    const uint64_t* __restrict__ off2, const uint32_t* __restrict__ dst2,
    const uint32_t* __restrict__ rows, uint64_t nrows, uint32_t v_begin,
    uint32_t* __restrict__ D) {
  __shared__ uint32_t s_lab[4][kCdlpWaveSlots];
  __shared__ uint32_t s_distinct[4];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  uint64_t groups = (nrows + 3) / 4;
  for (uint64_t grp = blockIdx.x; grp < groups; grp += gridDim.x) {
    uint64_t i = grp * 4 + wid;
    const bool have = i < nrows;
    uint32_t r = have ? rows[i] : 0;
    uint32_t v = v_begin + r;
    uint32_t d1 = 0, deg = 0, cap = 64, mask = 63;
    uint64_t b1 = 0, b2 = 0;
    if (have) {
      b1 = off1[r];
      d1 = static_cast<uint32_t>(off1[r + 1] - b1);
      deg = d1;
      if (off2) {
        b2 = off2[r];
        deg += static_cast<uint32_t>(off2[r + 1] - b2);
      }
      while (cap < 2 * deg) cap <<= 1;
      if (cap > kCdlpWaveSlots) cap = kCdlpWaveSlots;
      mask = cap - 1;
      for (uint32_t k = lane; k < cap; k += kWave)
        s_lab[wid][k] = kCdlpEmpty;
      if (lane == 0) s_distinct[wid] = 0;
    }
    __syncthreads();
    if (have) {
      for (uint32_t k = lane; k < deg; k += kWave) {
        uint32_t u = k < d1 ? dst1[b1 + k] : dst2[b2 + (k - d1)];
        if (u == v) continue;
        uint32_t idx = cdlp_hash(u) & mask;
        for (;;) {
          uint32_t old = atomicCAS(&s_lab[wid][idx], kCdlpEmpty, u);
          if (old == kCdlpEmpty) {
            atomicAdd(&s_distinct[wid], 1u);
            break;
          }
          if (old == u) break;
          idx = (idx + 1) & mask;
        }
      }
    }
    __syncthreads();
    if (have && lane == 0) D[v] = s_distinct[wid];
    __syncthreads();
  }
}

__global__ void lcc_orient_wave_kernel(
    const uint64_t* __restrict__ off1, const uint32_t* __restrict__ dst1,
    const uint64_t* __restrict__ off2, const uint32_t* __restrict__ dst2,
    const uint32_t* __restrict__ rows, uint64_t nrows, uint32_t v_begin,
    const uint32_t* __restrict__ D, const uint64_t* __restrict__ ooff,
    uint32_t* __restrict__ oadj, uint32_t* __restrict__ ocnt, bool filter) {
  __shared__ uint32_t s_lab[4][kCdlpWaveSlots];
  __shared__ uint32_t s_cursor[4];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  uint64_t groups = (nrows + 3) / 4;
  for (uint64_t grp = blockIdx.x; grp < groups; grp += gridDim.x) {
    uint64_t i = grp * 4 + wid;
    const bool have = i < nrows;
    uint32_t r = have ? rows[i] : 0;
    uint32_t v = v_begin + r;
    uint32_t d1 = 0, deg = 0, cap = 64, mask = 63;
    uint64_t b1 = 0, b2 = 0;
    if (have) {
      b1 = off1[r];
      d1 = static_cast<uint32_t>(off1[r + 1] - b1);
      deg = d1;
      if (off2) {
        b2 = off2[r];
        deg += static_cast<uint32_t>(off2[r + 1] - b2);
      }
      while (cap < 2 * deg) cap <<= 1;
      if (cap > kCdlpWaveSlots) cap = kCdlpWaveSlots;
      mask = cap - 1;
      for (uint32_t k = lane; k < cap; k += kWave)
        s_lab[wid][k] = kCdlpEmpty;
      if (lane == 0) s_cursor[wid] = 0;
    }
    __syncthreads();
    if (have) {
      uint32_t dv = D[v];
      uint64_t base = ooff[r];
      for (uint32_t k = lane; k < deg; k += kWave) {
        uint32_t u = k < d1 ? dst1[b1 + k] : dst2[b2 + (k - d1)];
        if (u == v) continue;
        uint32_t idx = cdlp_hash(u) & mask;
        for (;;) {
          uint32_t old = atomicCAS(&s_lab[wid][idx], kCdlpEmpty, u);
          if (old == kCdlpEmpty) {
            if (!filter || lcc_keep(D[u], u, dv, v))
              oadj[base + atomicAdd(&s_cursor[wid], 1u)] = u;
            break;
          }
          if (old == u) break;
          idx = (idx + 1) & mask;
        }
      }
    }
    __syncthreads();
    if (have && lane == 0) ocnt[r] = s_cursor[wid];
    __syncthreads();
  }
}

// --- pass 2: build oriented adjacency (dedup + orientation filter) --------

__global__ void lcc_orient_small_kernel(
    const uint64_t* __restrict__ off1, const uint32_t* __restrict__ dst1,
    const uint64_t* __restrict__ off2, const uint32_t* __restrict__ dst2,
    const uint32_t* __restrict__ rows, uint64_t nrows, uint32_t v_begin,
    const uint32_t* __restrict__ D, const uint64_t* __restrict__ ooff,
    uint32_t* __restrict__ oadj, uint32_t* __restrict__ ocnt, bool filter) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wpb = kBlock / kWave;
  size_t wstride = static_cast<size_t>(gridDim.x) * wpb;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * wpb + wid; i < nrows;
       i += wstride) {
    uint32_t r = rows[i];
    uint32_t v = v_begin + r;
    uint64_t b1 = off1[r], e1 = off1[r + 1];
    uint32_t d1 = static_cast<uint32_t>(e1 - b1);
    uint32_t deg = d1;
    uint64_t b2 = 0;
    if (off2) {
      b2 = off2[r];
      deg += static_cast<uint32_t>(off2[r + 1] - b2);
    }
    uint32_t u = kCdlpEmpty;
    if (lane < static_cast<int>(deg))
      u = lane < static_cast<int>(d1) ? dst1[b1 + lane]
                                      : dst2[b2 + (lane - d1)];
    bool active = lane < static_cast<int>(deg) && u != v;
    // first-occurrence mask (wave dedup): the lowest lane holding each
    // distinct value is its representative
    unsigned long long remaining = __ballot(active);
    unsigned long long first_mask = 0;
    while (remaining) {
      int L = __ffsll(remaining) - 1;
      uint32_t u0 = __shfl(u, L, 64);
      unsigned long long m = __ballot(active && u == u0);
      first_mask |= (1ull << L);
      remaining &= ~m;
    }
    uint32_t dv = D[v];
    bool keep = active && ((first_mask >> lane) & 1) &&
                (!filter || lcc_keep(D[u], u, dv, v));
    unsigned long long keep_mask = __ballot(keep);
    uint32_t pos = __popcll(keep_mask & ((1ull << lane) - 1));
    uint64_t base = ooff[r];
    if (keep) oadj[base + pos] = u;
    if (lane == 0) ocnt[r] = __popcll(keep_mask);
  }
}

__global__ void lcc_orient_mid_kernel(
    const uint64_t* __restrict__ off1, const uint32_t* __restrict__ dst1,
    const uint64_t* __restrict__ off2, const uint32_t* __restrict__ dst2,
    const uint32_t* __restrict__ rows, uint64_t nrows, uint32_t v_begin,
    const uint32_t* __restrict__ D, const uint64_t* __restrict__ ooff,
    uint32_t* __restrict__ oadj, uint32_t* __restrict__ ocnt, bool filter) {
  __shared__ uint32_t s_lab[kLccLdsSlots];
  __shared__ uint32_t s_cursor;
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t r = rows[i];
    uint32_t v = v_begin + r;
    uint64_t b1 = off1[r], e1 = off1[r + 1];
    uint32_t d1 = static_cast<uint32_t>(e1 - b1);
    uint32_t deg = d1;
    uint64_t b2 = 0;
    if (off2) {
      b2 = off2[r];
      deg += static_cast<uint32_t>(off2[r + 1] - b2);
    }
    uint32_t cap = 64;
    while (cap < 2 * deg) cap <<= 1;
    if (cap > kLccLdsSlots) cap = kLccLdsSlots;
    const uint32_t mask = cap - 1;
    for (uint32_t k = threadIdx.x; k < cap; k += blockDim.x)
      s_lab[k] = kCdlpEmpty;
    if (threadIdx.x == 0) s_cursor = 0;
    __syncthreads();
    uint32_t dv = D[v];
    uint64_t base = ooff[r];
    for (uint32_t k = threadIdx.x; k < deg; k += blockDim.x) {
      uint32_t u = k < d1 ? dst1[b1 + k] : dst2[b2 + (k - d1)];
      if (u == v) continue;
      uint32_t idx = cdlp_hash(u) & mask;
      for (;;) {
        uint32_t old = atomicCAS(&s_lab[idx], kCdlpEmpty, u);
        if (old == kCdlpEmpty) {
          if (!filter || lcc_keep(D[u], u, dv, v))
            oadj[base + atomicAdd(&s_cursor, 1u)] = u;
          break;
        }
        if (old == u) break;
        idx = (idx + 1) & mask;
      }
    }
    __syncthreads();
    if (threadIdx.x == 0) ocnt[r] = s_cursor;
    __syncthreads();
  }
}

__global__ void lcc_orient_large_kernel(
    const uint64_t* __restrict__ off1, const uint32_t* __restrict__ dst1,
    const uint64_t* __restrict__ off2, const uint32_t* __restrict__ dst2,
    const uint32_t* __restrict__ rows, uint64_t nrows,
    const uint64_t* __restrict__ tbl_off, uint32_t* __restrict__ tbl_lab,
    uint32_t v_begin, const uint32_t* __restrict__ D,
    const uint64_t* __restrict__ ooff, uint32_t* __restrict__ oadj,
    uint32_t* __restrict__ ocnt, bool filter) {
  __shared__ uint32_t s_cursor;
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t r = rows[i];
    uint32_t v = v_begin + r;
    uint64_t b1 = off1[r], e1 = off1[r + 1];
    uint64_t d1 = e1 - b1;
    uint64_t deg = d1;
    uint64_t b2 = 0;
    if (off2) {
      b2 = off2[r];
      deg += off2[r + 1] - b2;
    }
    uint64_t tb = tbl_off[i];
    const uint64_t mask = (tbl_off[i + 1] - tb) - 1;
    uint32_t* tl = tbl_lab + tb;
    if (threadIdx.x == 0) s_cursor = 0;
    __syncthreads();
    uint32_t dv = D[v];
    uint64_t base = ooff[r];
    for (uint64_t k = threadIdx.x; k < deg; k += blockDim.x) {
      uint32_t u = k < d1 ? dst1[b1 + k] : dst2[b2 + (k - d1)];
      if (u == v) continue;
      uint64_t idx = cdlp_hash(u) & mask;
      for (;;) {
        uint32_t old = atomicCAS(&tl[idx], kCdlpEmpty, u);
        if (old == kCdlpEmpty) {
          if (!filter || lcc_keep(D[u], u, dv, v))
            oadj[base + atomicAdd(&s_cursor, 1u)] = u;
          break;
        }
        if (old == u) break;
        idx = (idx + 1) & mask;
      }
    }
    __syncthreads();
    if (threadIdx.x == 0) ocnt[r] = s_cursor;
    __syncthreads();
  }
}

// compact capacity-layout rows into the tight global CSR region
__global__ void lcc_compact_kernel(const uint64_t* __restrict__ ooff,
                                   const uint32_t* __restrict__ ocnt,
                                   const uint32_t* __restrict__ oadj,
                                   const uint64_t* __restrict__ goff,
                                   uint32_t owned, uint32_t v_begin,
                                   uint32_t* __restrict__ gdst) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wpb = kBlock / kWave;
  size_t wstride = static_cast<size_t>(gridDim.x) * wpb;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * wpb + wid; i < owned;
       i += wstride) {
    uint32_t n = ocnt[i];
    uint64_t src = ooff[i];
    uint64_t dstb = goff[v_begin + i];
    for (uint32_t k = lane; k < n; k += kWave) gdst[dstb + k] = oadj[src + k];
  }
}

// bucket rows for the sort tiers by oriented count (block-aggregated
// counter reservation, same pattern as bucket_rows_kernel)
__global__ void lcc_sortbucket_kernel(const uint64_t* __restrict__ goff,
                                      uint32_t owned, uint32_t v_begin,
                                      uint32_t* lds_rows,
                                      unsigned long long* c_lds,
                                      uint32_t* big_rows,
                                      unsigned long long* c_big) {
  __shared__ uint32_t s_cnt[2];
  __shared__ unsigned long long s_base[2];
  uint32_t* lists[2] = {lds_rows, big_rows};
  unsigned long long* gcnt[2] = {c_lds, c_big};
  const uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t base = blockIdx.x * blockDim.x; base < owned;
       base += stride) {
    if (threadIdx.x < 2) s_cnt[threadIdx.x] = 0;
    __syncthreads();
    uint32_t r = base + threadIdx.x;
    int b = -1;
    uint32_t loc = 0;
    if (r < owned) {
      uint64_t n = goff[v_begin + r + 1] - goff[v_begin + r];
      if (n >= 2) {
        b = n <= kLccSortLds ? 0 : 1;
        loc = atomicAdd(&s_cnt[b], 1u);
      }
    }
    __syncthreads();
    if (threadIdx.x < 2 && s_cnt[threadIdx.x])
      s_base[threadIdx.x] =
          atomicAdd(gcnt[threadIdx.x],
                    static_cast<unsigned long long>(s_cnt[threadIdx.x]));
    __syncthreads();
    if (b >= 0) lists[b][s_base[b] + loc] = r;
    __syncthreads();
  }
}

__device__ __forceinline__ void bitonic_stage(uint32_t* a, uint32_t n,
                                              uint32_t k, uint32_t j) {
  for (uint32_t i = threadIdx.x; i < n; i += blockDim.x) {
    uint32_t ij = i ^ j;
    if (ij > i) {
      bool up = (i & k) == 0;
      uint32_t x = a[i], y = a[ij];
      if ((x > y) == up) {
        a[i] = y;
        a[ij] = x;
      }
    }
  }
}

// LDS bitonic over rows with count <= kLccSortLds (pad with 0xFFFFFFFF)
__global__ void lcc_sort_lds_kernel(const uint64_t* __restrict__ goff,
                                    const uint32_t* __restrict__ rows,
                                    uint64_t nrows, uint32_t v_begin,
                                    uint32_t* __restrict__ gdst) {
  __shared__ uint32_t s_a[kLccSortLds];
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t r = rows[i];
    uint64_t b = goff[v_begin + r];
    uint32_t n = static_cast<uint32_t>(goff[v_begin + r + 1] - b);
    uint32_t cap = 2;
    while (cap < n) cap <<= 1;
    for (uint32_t k = threadIdx.x; k < cap; k += blockDim.x)
      s_a[k] = k < n ? gdst[b + k] : 0xFFFFFFFFu;
    __syncthreads();
    for (uint32_t k = 2; k <= cap; k <<= 1)
      for (uint32_t j = k >> 1; j > 0; j >>= 1) {
        bitonic_stage(s_a, cap, k, j);
        __syncthreads();
      }
    for (uint32_t k = threadIdx.x; k < n; k += blockDim.x) gdst[b + k] = s_a[k];
    __syncthreads();
  }
}

// global-scratch bitonic for heavy rows (pow2-padded pool)
__global__ void lcc_bigpad_kernel(const uint64_t* __restrict__ goff,
                                  const uint32_t* __restrict__ rows,
                                  uint64_t nrows,
                                  const uint64_t* __restrict__ pad_off,
                                  const uint32_t* __restrict__ gdst,
                                  uint32_t v_begin,
                                  uint32_t* __restrict__ scratch) {
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t r = rows[i];
    uint64_t b = goff[v_begin + r];
    uint64_t n = goff[v_begin + r + 1] - b;
    uint64_t pb = pad_off[i];
    uint64_t cap = pad_off[i + 1] - pb;
    for (uint64_t k = threadIdx.x; k < cap; k += blockDim.x)
      scratch[pb + k] = k < n ? gdst[b + k] : 0xFFFFFFFFu;
  }
}

__global__ void lcc_sort_big_kernel(const uint64_t* __restrict__ pad_off,
                                    uint64_t nrows,
                                    uint32_t* __restrict__ scratch) {
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint64_t pb = pad_off[i];
    uint32_t cap = static_cast<uint32_t>(pad_off[i + 1] - pb);
    uint32_t* a = scratch + pb;
    for (uint32_t k = 2; k <= cap; k <<= 1)
      for (uint32_t j = k >> 1; j > 0; j >>= 1) {
        bitonic_stage(a, cap, k, j);
        __syncthreads();
      }
  }
}

__global__ void lcc_bigunpad_kernel(const uint64_t* __restrict__ goff,
                                    const uint32_t* __restrict__ rows,
                                    uint64_t nrows,
                                    const uint64_t* __restrict__ pad_off,
                                    const uint32_t* __restrict__ scratch,
                                    uint32_t v_begin,
                                    uint32_t* __restrict__ gdst) {
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t r = rows[i];
    uint64_t b = goff[v_begin + r];
    uint64_t n = goff[v_begin + r + 1] - b;
    uint64_t pb = pad_off[i];
    for (uint64_t k = threadIdx.x; k < n; k += blockDim.x)
      gdst[b + k] = scratch[pb + k];
  }
}

__global__ void lcc_bigcap_kernel(const uint64_t* __restrict__ goff,
                                  const uint32_t* __restrict__ rows,
                                  uint64_t nrows, uint32_t v_begin,
                                  uint32_t* __restrict__ caps) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * blockDim.x +
                    threadIdx.x;
       i < nrows; i += stride) {
    uint32_t r = rows[i];
    uint64_t n = goff[v_begin + r + 1] - goff[v_begin + r];
    uint64_t cap = 2;
    while (cap < n) cap <<= 1;
    caps[i] = static_cast<uint32_t>(cap);
  }
}

// Triangle counting via per-vertex hash sets of oriented neighbors.
// v2 history: per-lane serial merge rescanned O(u) per edge (quadratic in
// hub rows); binary search made it WORSE (log n random reads beat by the
// merge's streaming locality — 2.3s -> 3.7s at 40M). Hash probing is O(1)
// random reads per element: enumerate the smaller list, probe the larger
// list's open-addressing set. Tables are built once over the exchanged
// oriented CSR (capacity 2x, pow2), ~8 bytes per oriented edge of HBM.
__global__ void lcc_hashcap_kernel(const uint64_t* __restrict__ goff,
                                   uint32_t nv, uint32_t* __restrict__ caps) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t v = blockIdx.x * blockDim.x + threadIdx.x; v < nv;
       v += stride) {
    uint64_t n = goff[v + 1] - goff[v];
    uint64_t cap = 1;  // empty rows are never probed
    if (n) {
      cap = 4;
      while (cap < 2 * n) cap <<= 1;
    }
    caps[v] = static_cast<uint32_t>(cap);
  }
}

__global__ void lcc_hashfill_kernel(const uint64_t* __restrict__ goff,
                                    const uint32_t* __restrict__ gdst,
                                    const uint64_t* __restrict__ hoff,
                                    uint32_t nv,
                                    uint32_t* __restrict__ htab) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wpb = kBlock / kWave;
  size_t wstride = static_cast<size_t>(gridDim.x) * wpb;
  for (uint64_t v = static_cast<uint64_t>(blockIdx.x) * wpb + wid; v < nv;
       v += wstride) {
    uint64_t b = goff[v];
    uint32_t n = static_cast<uint32_t>(goff[v + 1] - b);
    if (n == 0) continue;
    uint64_t hb = hoff[v];
    const uint64_t mask = (hoff[v + 1] - hb) - 1;
    uint32_t* t = htab + hb;
    for (uint32_t k = lane; k < n; k += kWave) {
      uint32_t x = gdst[b + k];
      uint64_t idx = cdlp_hash(x) & mask;
      for (;;) {
        uint32_t old = atomicCAS(&t[idx], kCdlpEmpty, x);
        if (old == kCdlpEmpty || old == x) break;
        idx = (idx + 1) & mask;
      }
    }
  }
}

__device__ __forceinline__ bool lcc_probe(const uint32_t* __restrict__ t,
                                          uint64_t mask, uint32_t key) {
  uint64_t idx = cdlp_hash(key) & mask;
  for (;;) {
    uint32_t x = t[idx];
    if (x == key) return true;
    if (x == kCdlpEmpty) return false;
    idx = (idx + 1) & mask;
  }
}

// Light-edge pass, two tiers by oriented row size:
//   wee rows (un <= 32): wave-per-row, lanes probe global hash sets
//   staged rows: block-per-row; the row's hash set is staged into LDS once
//   and every thread's probes hit LDS (the global-probe version spent its
//   time on scattered 4B HBM reads; witness atomics measured ~5%)
constexpr uint32_t kLccWeeRows = 768;  // swept 32/256/768/2048: 3206/2750/2607/2713 ms at datagen-9_0
constexpr uint32_t kLccStageSlots = 8192;  // 32 KB LDS

__global__ void lcc_tri_wee_kernel(const uint64_t* __restrict__ goff,
                                   const uint32_t* __restrict__ gdst,
                                   const uint64_t* __restrict__ hoff,
                                   const uint32_t* __restrict__ htab,
                                   const uint32_t* __restrict__ rows,
                                   uint64_t nrows, uint32_t v_begin,
                                   unsigned long long* __restrict__ T,
                                   uint32_t heavy_thresh,
                                   unsigned long long* __restrict__ heavy_q,
                                   unsigned long long* __restrict__ heavy_n,
                                   bool skip_witness) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wpb = kBlock / kWave;
  size_t wstride = static_cast<size_t>(gridDim.x) * wpb;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * wpb + wid; i < nrows;
       i += wstride) {
    uint32_t u = v_begin + rows[i];
    uint64_t ub = goff[u];
    uint32_t un = static_cast<uint32_t>(goff[u + 1] - ub);
    unsigned long long my_u = 0;
    for (uint32_t k = lane; k < un; k += kWave) {
      uint32_t v = gdst[ub + k];
      uint64_t vb = goff[v];
      uint32_t vn = static_cast<uint32_t>(goff[v + 1] - vb);
      if (vn == 0) continue;
      uint32_t mn = vn < un ? vn : un;
      if (mn > heavy_thresh) {
        heavy_q[atomicAdd(heavy_n, 1ull)] =
            (static_cast<unsigned long long>(u) << 32) | v;
        continue;
      }
      uint64_t eb, hb;
      uint32_t en;
      uint64_t mask;
      if (vn < un) {
        eb = vb;
        en = vn;
        hb = hoff[u];
        mask = (hoff[u + 1] - hb) - 1;
      } else {
        eb = ub;
        en = un;
        hb = hoff[v];
        mask = (hoff[v + 1] - hb) - 1;
      }
      const uint32_t* t = htab + hb;
      unsigned long long hits = 0;
      // 4 interleaved probe chains (see the heavy kernel): the serial
      // dependent probe left the wave parked on latency
      for (uint32_t e0 = 0; e0 < en; e0 += 4) {
        uint32_t key[4];
        uint64_t idx[4];
        bool live[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          live[j] = e0 + j < en;
          key[j] = live[j] ? gdst[eb + e0 + j] : 0;
          idx[j] = cdlp_hash(key[j]) & mask;
        }
        for (;;) {
          uint32_t x[4];
#pragma unroll
          for (int j = 0; j < 4; ++j)
            if (live[j]) x[j] = t[idx[j]];
          bool any = false;
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            if (!live[j]) continue;
            if (x[j] == key[j]) {
              ++hits;
              if (!skip_witness) atomicAdd(&T[key[j]], 1ull);
              live[j] = false;
            } else if (x[j] == kCdlpEmpty) {
              live[j] = false;
            } else {
              idx[j] = (idx[j] + 1) & mask;
              any = true;
            }
          }
          if (!any) break;
        }
      }
      if (hits) {
        my_u += hits;
        atomicAdd(&T[v], hits);
      }
    }
#pragma unroll
    for (int d = 32; d > 0; d >>= 1)
      my_u += __shfl_down(my_u, d, 64);
    if (lane == 0 && my_u) atomicAdd(&T[u], my_u);
  }
}

__device__ __forceinline__ bool lcc_probe_lds(
    const uint32_t* __restrict__ t, uint32_t mask, uint32_t key) {
  uint32_t idx = cdlp_hash(key) & mask;
  for (;;) {
    uint32_t x = t[idx];
    if (x == key) return true;
    if (x == kCdlpEmpty) return false;
    idx = (idx + 1) & mask;
  }
}

__global__ void lcc_tri_staged_kernel(
    const uint64_t* __restrict__ goff, const uint32_t* __restrict__ gdst,
    const uint64_t* __restrict__ hoff, const uint32_t* __restrict__ htab,
    const uint32_t* __restrict__ rows, uint64_t nrows, uint32_t v_begin,
    unsigned long long* __restrict__ T, uint32_t heavy_thresh,
    unsigned long long* __restrict__ heavy_q,
    unsigned long long* __restrict__ heavy_n, bool skip_witness) {
  __shared__ uint32_t s_tab[kLccStageSlots];
  __shared__ unsigned long long s_hits;
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t u = v_begin + rows[i];
    uint64_t ub = goff[u];
    uint32_t un = static_cast<uint32_t>(goff[u + 1] - ub);
    uint64_t hb = hoff[u];
    uint32_t ucap = static_cast<uint32_t>(hoff[u + 1] - hb);
    const bool stage = ucap <= kLccStageSlots;
    if (stage)
      for (uint32_t k = threadIdx.x; k < ucap; k += blockDim.x)
        s_tab[k] = htab[hb + k];
    if (threadIdx.x == 0) s_hits = 0;
    __syncthreads();
    const uint32_t umask = ucap - 1;
    const uint32_t* ut = htab + hb;
    unsigned long long my_u = 0;
    for (uint32_t k = threadIdx.x; k < un; k += blockDim.x) {
      uint32_t v = gdst[ub + k];
      uint64_t vb = goff[v];
      uint32_t vn = static_cast<uint32_t>(goff[v + 1] - vb);
      if (vn == 0) continue;
      uint32_t mn = vn < un ? vn : un;
      if (mn > heavy_thresh) {
        heavy_q[atomicAdd(heavy_n, 1ull)] =
            (static_cast<unsigned long long>(u) << 32) | v;
        continue;
      }
      unsigned long long hits = 0;
      if (vn < un) {
        // enumerate O(v), probe u's set (LDS when staged)
        for (uint32_t e = 0; e < vn; ++e) {
          uint32_t w = gdst[vb + e];
          bool hit = stage ? lcc_probe_lds(s_tab, umask, w)
                           : lcc_probe(ut, umask, w);
          if (hit) {
            ++hits;
            if (!skip_witness) atomicAdd(&T[w], 1ull);
          }
        }
      } else {
        // enumerate O(u) (streamed, shared), probe v's global set
        uint64_t vhb = hoff[v];
        const uint64_t vmask = (hoff[v + 1] - vhb) - 1;
        const uint32_t* vt = htab + vhb;
        for (uint32_t e = 0; e < un; ++e) {
          uint32_t w = gdst[ub + e];
          if (lcc_probe(vt, vmask, w)) {
            ++hits;
            if (!skip_witness) atomicAdd(&T[w], 1ull);
          }
        }
      }
      if (hits) {
        my_u += hits;
        atomicAdd(&T[v], hits);
      }
    }
    if (my_u) atomicAdd(&s_hits, my_u);
    __syncthreads();
    if (threadIdx.x == 0 && s_hits) atomicAdd(&T[u], s_hits);
    __syncthreads();
  }
}

// bucket rows for the two light tiers (skips empty rows)
__global__ void lcc_tribucket_kernel(const uint64_t* __restrict__ goff,
                                     uint32_t owned, uint32_t v_begin,
                                     uint32_t* wee, unsigned long long* cw,
                                     uint32_t* big, unsigned long long* cb) {
  __shared__ uint32_t s_cnt[2];
  __shared__ unsigned long long s_base[2];
  uint32_t* lists[2] = {wee, big};
  unsigned long long* gcnt[2] = {cw, cb};
  const uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t base = blockIdx.x * blockDim.x; base < owned;
       base += stride) {
    if (threadIdx.x < 2) s_cnt[threadIdx.x] = 0;
    __syncthreads();
    uint32_t r = base + threadIdx.x;
    int b = -1;
    uint32_t loc = 0;
    if (r < owned) {
      uint64_t n = goff[v_begin + r + 1] - goff[v_begin + r];
      if (n) {
        b = n <= kLccWeeRows ? 0 : 1;
        loc = atomicAdd(&s_cnt[b], 1u);
      }
    }
    __syncthreads();
    if (threadIdx.x < 2 && s_cnt[threadIdx.x])
      s_base[threadIdx.x] =
          atomicAdd(gcnt[threadIdx.x],
                    static_cast<unsigned long long>(s_cnt[threadIdx.x]));
    __syncthreads();
    if (b >= 0) lists[b][s_base[b] + loc] = r;
    __syncthreads();
  }
}

// second pass: one wave per heavy edge; lanes stride the smaller list
__global__ void lcc_triangle_heavy_kernel(
    const uint64_t* __restrict__ goff, const uint32_t* __restrict__ gdst,
    const uint64_t* __restrict__ hoff, const uint32_t* __restrict__ htab,
    const unsigned long long* __restrict__ heavy_q, uint64_t heavy_n,
    unsigned long long* __restrict__ T) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wpb = kBlock / kWave;
  size_t wstride = static_cast<size_t>(gridDim.x) * wpb;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * wpb + wid;
       i < heavy_n; i += wstride) {
    unsigned long long pk = heavy_q[i];
    uint32_t u = static_cast<uint32_t>(pk >> 32);
    uint32_t v = static_cast<uint32_t>(pk);
    uint64_t ub = goff[u];
    uint32_t un = static_cast<uint32_t>(goff[u + 1] - ub);
    uint64_t vb = goff[v];
    uint32_t vn = static_cast<uint32_t>(goff[v + 1] - vb);
    uint64_t eb, hb_big, hcap_end;
    uint32_t en;
    if (vn < un) {
      eb = vb;
      en = vn;
      hb_big = hoff[u];
      hcap_end = hoff[u + 1];
    } else {
      eb = ub;
      en = un;
      hb_big = hoff[v];
      hcap_end = hoff[v + 1];
    }
    const uint64_t mask = (hcap_end - hb_big) - 1;
    const uint32_t* t = htab + hb_big;
    unsigned long long hits = 0;
    // four probe state machines per lane: the single dependent probe chain
    // left waves 93% parked on memory latency (PMC r01); 4 outstanding
    // loads per lane quadruples the memory-level parallelism
    for (uint32_t e0 = lane; e0 < en; e0 += kWave * 4) {
      uint32_t key[4];
      uint64_t idx[4];
      bool live[4];
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        uint32_t e = e0 + j * kWave;
        live[j] = e < en;
        key[j] = live[j] ? gdst[eb + e] : 0;
        idx[j] = cdlp_hash(key[j]) & mask;
      }
      for (;;) {
        uint32_t x[4];
#pragma unroll
        for (int j = 0; j < 4; ++j)
          if (live[j]) x[j] = t[idx[j]];
        bool any = false;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          if (!live[j]) continue;
          if (x[j] == key[j]) {
            ++hits;
            atomicAdd(&T[key[j]], 1ull);
            live[j] = false;
          } else if (x[j] == kCdlpEmpty) {
            live[j] = false;
          } else {
            idx[j] = (idx[j] + 1) & mask;
            any = true;
          }
        }
        if (!any) break;
      }
    }
#pragma unroll
    for (int d = 32; d > 0; d >>= 1)
      hits += __shfl_down(hits, d, 64);
    if (lane == 0 && hits) {
      atomicAdd(&T[u], hits);
      atomicAdd(&T[v], hits);
    }
  }
}

// --- directed LCC (reference cuda/lcc/lcc_directed*.h semantics) ---------
// numerator(v) = sum over u in N(v) of |N(v) ∩ Nout(u)| computed directly
// (no orientation trick applies to the directed definition). Same light
// tiers as the undirected counter: wee rows wave-per-row with global
// probes; bigger rows block-per-row with U(v)'s hash set staged in LDS.
__global__ void lcc_dir_bucket_kernel(const uint64_t* __restrict__ goffU,
                                      uint32_t owned, uint32_t v_begin,
                                      uint32_t* wee, unsigned long long* cw,
                                      uint32_t* big,
                                      unsigned long long* cb) {
  __shared__ uint32_t s_cnt[2];
  __shared__ unsigned long long s_base[2];
  uint32_t* lists[2] = {wee, big};
  unsigned long long* gcnt[2] = {cw, cb};
  const uint32_t stride = gridDim.x * blockDim.x;
  for (uint32_t base = blockIdx.x * blockDim.x; base < owned;
       base += stride) {
    if (threadIdx.x < 2) s_cnt[threadIdx.x] = 0;
    __syncthreads();
    uint32_t r = base + threadIdx.x;
    int b = -1;
    uint32_t loc = 0;
    if (r < owned) {
      uint64_t n = goffU[v_begin + r + 1] - goffU[v_begin + r];
      if (n >= 2) {
        b = n <= kLccWeeRows ? 0 : 1;
        loc = atomicAdd(&s_cnt[b], 1u);
      }
    }
    __syncthreads();
    if (threadIdx.x < 2 && s_cnt[threadIdx.x])
      s_base[threadIdx.x] =
          atomicAdd(gcnt[threadIdx.x],
                    static_cast<unsigned long long>(s_cnt[threadIdx.x]));
    __syncthreads();
    if (b >= 0) lists[b][s_base[b] + loc] = r;
    __syncthreads();
  }
}

__global__ void lcc_dir_count_wee_kernel(
    const uint64_t* __restrict__ goffU, const uint32_t* __restrict__ gdstU,
    const uint64_t* __restrict__ hoffU, const uint32_t* __restrict__ htabU,
    const uint64_t* __restrict__ goffO, const uint32_t* __restrict__ gdstO,
    const uint64_t* __restrict__ hoffO, const uint32_t* __restrict__ htabO,
    const uint32_t* __restrict__ rows, uint64_t nrows, uint32_t v_begin,
    unsigned long long* __restrict__ T, uint32_t heavy_thresh,
    unsigned long long* __restrict__ heavy_q,
    unsigned long long* __restrict__ heavy_n) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wpb = kBlock / kWave;
  size_t wstride = static_cast<size_t>(gridDim.x) * wpb;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * wpb + wid; i < nrows;
       i += wstride) {
    uint32_t v = v_begin + rows[i];
    uint64_t ub = goffU[v];
    uint32_t un = static_cast<uint32_t>(goffU[v + 1] - ub);
    unsigned long long my_v = 0;
    for (uint32_t k = lane; k < un; k += kWave) {
      uint32_t u = gdstU[ub + k];
      uint64_t ob = goffO[u];
      uint32_t on = static_cast<uint32_t>(goffO[u + 1] - ob);
      if (on == 0) continue;
      uint32_t mn = on < un ? on : un;
      if (mn > heavy_thresh) {
        heavy_q[atomicAdd(heavy_n, 1ull)] =
            (static_cast<unsigned long long>(v) << 32) | u;
        continue;
      }
      unsigned long long hits = 0;
      if (on <= un) {
        const uint64_t mask = (hoffU[v + 1] - hoffU[v]) - 1;
        const uint32_t* t = htabU + hoffU[v];
        for (uint32_t e = 0; e < on; ++e)
          if (lcc_probe(t, mask, gdstO[ob + e])) ++hits;
      } else {
        const uint64_t mask = (hoffO[u + 1] - hoffO[u]) - 1;
        const uint32_t* t = htabO + hoffO[u];
        for (uint32_t e = 0; e < un; ++e)
          if (lcc_probe(t, mask, gdstU[ub + e])) ++hits;
      }
      my_v += hits;
    }
#pragma unroll
    for (int d = 32; d > 0; d >>= 1)
      my_v += __shfl_down(my_v, d, 64);
    if (lane == 0 && my_v) atomicAdd(&T[v], my_v);
  }
}

__global__ void lcc_dir_count_staged_kernel(
    const uint64_t* __restrict__ goffU, const uint32_t* __restrict__ gdstU,
    const uint64_t* __restrict__ hoffU, const uint32_t* __restrict__ htabU,
    const uint64_t* __restrict__ goffO, const uint32_t* __restrict__ gdstO,
    const uint64_t* __restrict__ hoffO, const uint32_t* __restrict__ htabO,
    const uint32_t* __restrict__ rows, uint64_t nrows, uint32_t v_begin,
    unsigned long long* __restrict__ T, uint32_t heavy_thresh,
    unsigned long long* __restrict__ heavy_q,
    unsigned long long* __restrict__ heavy_n) {
  __shared__ uint32_t s_tab[kLccStageSlots];
  __shared__ unsigned long long s_hits;
  for (uint64_t i = blockIdx.x; i < nrows; i += gridDim.x) {
    uint32_t v = v_begin + rows[i];
    uint64_t ub = goffU[v];
    uint32_t un = static_cast<uint32_t>(goffU[v + 1] - ub);
    uint64_t hb = hoffU[v];
    uint32_t ucap = static_cast<uint32_t>(hoffU[v + 1] - hb);
    const bool stage = ucap <= kLccStageSlots;
    if (stage)
      for (uint32_t k = threadIdx.x; k < ucap; k += blockDim.x)
        s_tab[k] = htabU[hb + k];
    if (threadIdx.x == 0) s_hits = 0;
    __syncthreads();
    const uint32_t umask = ucap - 1;
    const uint32_t* ut = htabU + hb;
    unsigned long long my_v = 0;
    for (uint32_t k = threadIdx.x; k < un; k += blockDim.x) {
      uint32_t u = gdstU[ub + k];
      uint64_t ob = goffO[u];
      uint32_t on = static_cast<uint32_t>(goffO[u + 1] - ob);
      if (on == 0) continue;
      uint32_t mn = on < un ? on : un;
      if (mn > heavy_thresh) {
        heavy_q[atomicAdd(heavy_n, 1ull)] =
            (static_cast<unsigned long long>(v) << 32) | u;
        continue;
      }
      unsigned long long hits = 0;
      if (on <= un) {
        for (uint32_t e = 0; e < on; ++e) {
          uint32_t w = gdstO[ob + e];
          bool hit = stage ? lcc_probe_lds(s_tab, umask, w)
                           : lcc_probe(ut, umask, w);
          if (hit) ++hits;
        }
      } else {
        uint64_t ohb = hoffO[u];
        const uint64_t omask = (hoffO[u + 1] - ohb) - 1;
        const uint32_t* ot = htabO + ohb;
        for (uint32_t e = 0; e < un; ++e)
          if (lcc_probe(ot, omask, gdstU[ub + e])) ++hits;
      }
      my_v += hits;
    }
    if (my_v) atomicAdd(&s_hits, my_v);
    __syncthreads();
    if (threadIdx.x == 0 && s_hits) atomicAdd(&T[v], s_hits);
    __syncthreads();
  }
}

// heavy (wave-per-edge, lanes split the enumeration)
__global__ void lcc_dir_heavy_kernel(
    const uint64_t* __restrict__ goffU, const uint32_t* __restrict__ gdstU,
    const uint64_t* __restrict__ hoffU, const uint32_t* __restrict__ htabU,
    const uint64_t* __restrict__ goffO, const uint32_t* __restrict__ gdstO,
    const uint64_t* __restrict__ hoffO, const uint32_t* __restrict__ htabO,
    const unsigned long long* __restrict__ heavy_q, uint64_t heavy_n,
    unsigned long long* __restrict__ T) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wpb = kBlock / kWave;
  size_t wstride = static_cast<size_t>(gridDim.x) * wpb;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * wpb + wid;
       i < heavy_n; i += wstride) {
    unsigned long long pk = heavy_q[i];
    uint32_t v = static_cast<uint32_t>(pk >> 32);
    uint32_t u = static_cast<uint32_t>(pk);
    uint64_t ub = goffU[v];
    uint32_t un = static_cast<uint32_t>(goffU[v + 1] - ub);
    uint64_t ob = goffO[u];
    uint32_t on = static_cast<uint32_t>(goffO[u + 1] - ob);
    unsigned long long hits = 0;
    if (on <= un) {
      const uint64_t mask = (hoffU[v + 1] - hoffU[v]) - 1;
      const uint32_t* t = htabU + hoffU[v];
      for (uint32_t e = lane; e < on; e += kWave)
        if (lcc_probe(t, mask, gdstO[ob + e])) ++hits;
    } else {
      const uint64_t mask = (hoffO[u + 1] - hoffO[u]) - 1;
      const uint32_t* t = htabO + hoffO[u];
      for (uint32_t e = lane; e < un; e += kWave)
        if (lcc_probe(t, mask, gdstU[ub + e])) ++hits;
    }
#pragma unroll
    for (int d = 32; d > 0; d >>= 1)
      hits += __shfl_down(hits, d, 64);
    if (lane == 0 && hits) atomicAdd(&T[v], hits);
  }
}

// caps for the dedup'd OUT family: distinct-out <= min(out_deg, D)
__global__ void lcc_outcap_kernel(const uint64_t* __restrict__ off,
                                  const uint32_t* __restrict__ D,
                                  uint32_t owned, uint32_t v_begin,
                                  uint32_t* __restrict__ caps) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < owned;
       r += stride) {
    uint64_t d = off[r + 1] - off[r];
    uint32_t dd = D[v_begin + r];
    caps[r] = static_cast<uint32_t>(d < dd ? d : dd);
  }
}

__global__ void lcc_dir_finalize_kernel(
    const unsigned long long* __restrict__ T,
    const uint32_t* __restrict__ D, uint32_t owned, uint32_t v_begin,
    double* __restrict__ lcc) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < owned;
       r += stride) {
    uint32_t d = D[v_begin + r];
    lcc[r] = d < 2 ? 0.0
                   : static_cast<double>(T[v_begin + r]) /
                         (static_cast<double>(d) * (d - 1));
  }
}

__global__ void lcc_finalize_kernel(const unsigned long long* __restrict__ T,
                                    const uint32_t* __restrict__ D,
                                    uint32_t owned, uint32_t v_begin,
                                    double* __restrict__ lcc) {
  size_t stride = static_cast<size_t>(gridDim.x) * blockDim.x;
  for (uint32_t r = blockIdx.x * blockDim.x + threadIdx.x; r < owned;
       r += stride) {
    uint32_t d = D[v_begin + r];
    lcc[r] = d < 2 ? 0.0
                   : 2.0 * static_cast<double>(T[v_begin + r]) /
                         (static_cast<double>(d) * (d - 1));
  }
}

namespace {

// Shared piece: turn capacity-layout per-owned-row lists into a replicated
// global CSR (allgather counts -> scan -> compact -> per-rank broadcast).
// mark remote dsts of the capacity-layout oriented adjacency (wave/row)
__global__ void lcc_mark_oriented_kernel(
    const uint64_t* __restrict__ ooff, const uint32_t* __restrict__ ocnt,
    const uint32_t* __restrict__ oadj, uint32_t owned, uint32_t v_begin,
    uint32_t v_end, DevBitmap bm) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wpb = kBlock / kWave;
  size_t wstride = static_cast<size_t>(gridDim.x) * wpb;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * wpb + wid;
       i < owned; i += wstride) {
    uint64_t b = ooff[i];
    uint32_t n = ocnt[i];
    for (uint32_t k = lane; k < n; k += kWave) {
      uint32_t v = oadj[b + k];
      if (v < v_begin || v >= v_end) (void)bm.set_once(v);
    }
  }
}

// pack requested rows (sorted, variable length) into a contiguous payload
__global__ void lcc_pack_rows_kernel(const uint32_t* __restrict__ idx,
                                     uint64_t n,
                                     const uint64_t* __restrict__ goff,
                                     const uint32_t* __restrict__ gdst,
                                     const uint64_t* __restrict__ pack_off,
                                     uint32_t* __restrict__ out) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wpb = kBlock / kWave;
  size_t wstride = static_cast<size_t>(gridDim.x) * wpb;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * wpb + wid; i < n;
       i += wstride) {
    uint64_t b = goff[idx[i]];
    uint64_t o = pack_off[i], len = pack_off[i + 1] - o;
    for (uint64_t k = lane; k < len; k += kWave) out[o + k] = gdst[b + k];
  }
}

struct GlobalDedupCsr {
  DeviceBuffer<uint32_t> gcnt;
  DeviceBuffer<uint64_t> goff;
  DeviceBuffer<uint32_t> gdst;
  uint64_t total = 0;
};

// Fetch the referenced remote rows of a sparse-count CSR: each owner
// packs exactly the rows each peer requested (sidx/soff lists); payloads
// land directly in the requester's contiguous per-owner gdst regions
// (unfetched rows have zero count, so regions are hole-free). Counts
// must already be synced over the same lists.
void lcc_fetch_referenced_rows(GpuContext::Impl& I,
                               const DeviceBuffer<uint32_t>& sidx,
                               const std::vector<uint64_t>& soff,
                               const DeviceBuffer<uint32_t>& gcnt,
                               const DeviceBuffer<uint64_t>& goff,
                               DeviceBuffer<uint32_t>& gdst, uint32_t slice,
                               uint32_t nv_pad, int world, hipStream_t s) {
  uint64_t ns = soff[world];
  DeviceBuffer<uint32_t> slen(ns ? ns : 1);
  if (ns)
    gather4_kernel<<<grid_for(ns), kBlock, 0, s>>>(sidx.data(), ns,
                                                   gcnt.data(), slen.data());
  DeviceBuffer<uint64_t> spack(ns + 1);
  uint64_t stotal = exclusive_scan(slen.data(), spack.data(), ns, s, I.scan);
  DeviceBuffer<uint32_t> srows(stotal ? stotal : 1);
  if (ns)
    lcc_pack_rows_kernel<<<grid_for(ns * kWave), kBlock, 0, s>>>(
        sidx.data(), ns, goff.data(), gdst.data(), spack.data(),
        srows.data());
  // per-peer byte offsets: send = packed scan at list boundaries,
  // recv = goff at slice boundaries (both derive from the same counts)
  std::vector<uint64_t> sb(world + 1), rb(world + 1);
  for (int f = 0; f <= world; ++f) {
    uint64_t v = 0;
    HIP_CHECK(hipMemcpyAsync(&v, spack.data() + soff[f], 8,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    sb[f] = v * 4;
    uint64_t idx =
        std::min<uint64_t>(static_cast<uint64_t>(f) * slice, nv_pad);
    HIP_CHECK(hipMemcpyAsync(&v, goff.data() + idx, 8,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    rb[f] = v * 4;
  }
  I.dc.sendrecv(reinterpret_cast<const uint8_t*>(srows.data()), sb,
                reinterpret_cast<uint8_t*>(gdst.data()), rb, s);
  HIP_CHECK(hipStreamSynchronize(s));
}

// Compact a capacity-layout family into a sparse global-index CSR. With
// lists (multi): counts are synced over the request lists and referenced
// remote rows fetched (volume scales with the boundary). Without lists:
// own rows only — for families that are never probed at remote indices
// (directed U, read only at owned v).
void build_global_csr(GpuContext::Impl& I, int rank, int world,
                      uint32_t nv_pad, uint32_t slice, uint32_t owned,
                      uint32_t v_begin, const DeviceBuffer<uint64_t>& ooff,
                      const DeviceBuffer<uint32_t>& ocnt,
                      const DeviceBuffer<uint32_t>& oadj,
                      GlobalDedupCsr& out, hipStream_t s,
                      const DeviceBuffer<uint32_t>* ridx = nullptr,
                      const std::vector<uint64_t>* roff = nullptr,
                      const DeviceBuffer<uint32_t>* sidx = nullptr,
                      const std::vector<uint64_t>* soff = nullptr) {
  bool fetch = world > 1 && ridx;
  out.gcnt.resize(nv_pad);
  out.gcnt.zero(s);
  HIP_CHECK(hipMemcpyAsync(out.gcnt.data() + v_begin, ocnt.data(),
                           owned * 4, hipMemcpyDeviceToDevice, s));
  if (fetch) {
    DeviceBuffer<uint8_t> ss((*soff)[world] * 4 + 4),
        sr(roff->back() * 4 + 4);
    ref_sync_begin(I, *sidx, *soff, *roff, ss, sr, out.gcnt.data(), s);
    ref_sync_end(I, *ridx, *roff, sr, out.gcnt.data(), s);
  }
  out.goff.resize(static_cast<size_t>(nv_pad) + 1);
  out.total =
      exclusive_scan(out.gcnt.data(), out.goff.data(), nv_pad, s, I.scan);
  out.gdst.resize(out.total ? out.total : 1);
  if (owned)
    lcc_compact_kernel<<<grid_for(static_cast<size_t>(owned) * kWave),
                         kBlock, 0, s>>>(ooff.data(), ocnt.data(),
                                         oadj.data(), out.goff.data(),
                                         owned, v_begin, out.gdst.data());
  if (fetch)
    lcc_fetch_referenced_rows(I, *sidx, *soff, out.gcnt, out.goff,
                              out.gdst, slice, nv_pad, world, s);
  (void)rank;
}

void build_hash_sets(GpuContext::Impl& I, uint32_t nv_pad,
                     const GlobalDedupCsr& csr, DeviceBuffer<uint64_t>& hoff,
                     DeviceBuffer<uint32_t>& htab, hipStream_t s) {
  DeviceBuffer<uint32_t> hcap(nv_pad);
  hoff.resize(static_cast<size_t>(nv_pad) + 1);
  lcc_hashcap_kernel<<<grid_for(nv_pad), kBlock, 0, s>>>(csr.goff.data(),
                                                         nv_pad,
                                                         hcap.data());
  uint64_t total = exclusive_scan(hcap.data(), hoff.data(), nv_pad, s,
                                  I.scan);
  htab.resize(total);
  fill(htab.data(), kCdlpEmpty, total, s);
  lcc_hashfill_kernel<<<grid_for(static_cast<size_t>(nv_pad) * kWave),
                        kBlock, 0, s>>>(csr.goff.data(), csr.gdst.data(),
                                        hoff.data(), nv_pad, htab.data());
}

}  // namespace

// Directed LCC: build the union family U(v)=N(v) and the deduped OUT
// family Nout(v), then count |U(v) ∩ Nout(u)| per (v, u∈U(v)) pair.
GpuRunResult GpuContext::lcc_directed(DeviceGraph& g, bool fetch) {
  RangeMarker _mk("grapehip::lcc_directed");
  auto& I = *impl_;
  hipStream_t s = I.compute;
  uint32_t nv_pad = padded_nv(g, world_);
  uint32_t owned = g.owned();
  uint32_t slice = nv_pad / (world_ ? world_ : 1);
  bool multi = world_ > 1;
  if (!g.has_in)
    throw std::runtime_error(
        "GPU directed LCC needs the in-CSR (build_in_csr=True)");
  const uint64_t* off1 = g.oe_off.data();
  const uint32_t* dst1 = g.oe_dst.data();
  const uint64_t* off2 = g.ie_off.data();
  const uint32_t* dst2 = g.ie_dst.data();

  // referenced-row request lists (multi): only the O family is probed at
  // remote indices (kernels read U solely at owned v), so U stays local
  // and O rows are fetched for the u's own U rows actually name
  DeviceBuffer<uint32_t> ridx, sidx;
  std::vector<uint64_t> roff, soff;
  if (multi) ensure_mirrors(I, comm_, g, rank_, world_, s);

  if (comm_) comm_->barrier();
  HIP_CHECK(hipDeviceSynchronize());
  double t0 = wall_s();
  const uint64_t b_p2p0 = I.dc.bytes_p2p, b_coll0 = I.dc.bytes_coll;

  DeviceBuffer<uint32_t> Dv(nv_pad);
  Dv.zero(s);
  GlobalDedupCsr U, O;
  {
    DeviceBuffer<uint32_t> t_small(owned ? owned : 1),
        t_mid(owned ? owned : 1), t_large(owned ? owned : 1);
    DeviceBuffer<uint32_t> t_wavelds(owned ? owned : 1);
    DeviceBuffer<unsigned long long> cnts(5);
    cnts.zero(s);
    if (owned)
      cdlp_bucket_kernel<<<grid_for(owned), kBlock, 0, s>>>(
          off1, off2, owned, nullptr, cnts.data() + 3, t_small.data(),
          cnts.data() + 0, t_wavelds.data(), cnts.data() + 4, t_mid.data(),
          cnts.data() + 1, t_large.data(), cnts.data() + 2);
    auto hc = cnts.download(s);
    uint64_t n_small = hc[0], n_mid = hc[1], n_large = hc[2],
             n_wavelds = hc[4];

    DeviceBuffer<uint32_t> heavy_caps;
    DeviceBuffer<uint64_t> heavy_off;
    DeviceBuffer<uint32_t> tbl_lab;
    uint64_t heavy_total = 0;
    if (n_large) {
      heavy_caps.resize(n_large);
      heavy_off.resize(n_large + 1);
      cdlp_heavy_cap_kernel<<<grid_for(n_large), kBlock, 0, s>>>(
          off1, off2, t_large.data(), n_large, heavy_caps.data());
      heavy_total = exclusive_scan(heavy_caps.data(), heavy_off.data(),
                                   n_large, s, I.scan);
      tbl_lab.resize(heavy_total);
    }

    // distinct union counts
    if (n_small)
      lcc_distinct_small_kernel<<<grid_for(n_small * kWave), kBlock, 0,
                                  s>>>(off1, dst1, off2, dst2,
                                       t_small.data(), n_small, g.v_begin,
                                       Dv.data());
    if (n_wavelds)
      lcc_distinct_wave_kernel<<<grid_for(((n_wavelds + 3) / 4) * kBlock),
                                 kBlock, 0, s>>>(off1, dst1, off2, dst2,
                                                 t_wavelds.data(),
                                                 n_wavelds, g.v_begin,
                                                 Dv.data());
    if (n_mid)
      lcc_distinct_mid_kernel<<<std::min<int>(n_mid, kMaxGrid), kBlock, 0,
                                s>>>(off1, dst1, off2, dst2, t_mid.data(),
                                     n_mid, g.v_begin, Dv.data());
    if (n_large) {
      fill(tbl_lab.data(), kCdlpEmpty, heavy_total, s);
      lcc_distinct_large_kernel<<<std::min<int>(n_large, kMaxGrid), kBlock,
                                  0, s>>>(off1, dst1, off2, dst2,
                                          t_large.data(), n_large,
                                          heavy_off.data(), tbl_lab.data(),
                                          g.v_begin, Dv.data());
    }
    if (multi) {
      mirror_sync_begin(I, g, Dv.data(), s);
      mirror_sync_end(I, g, Dv.data(), s);
    }

    // family builder: run the dedup pass over (o1,d1,o2,d2) with given
    // capacities, then lift to a sparse global-index CSR (own rows, plus
    // fetched referenced rows when lists are supplied)
    auto build_family = [&](const uint64_t* o1, const uint32_t* d1,
                            const uint64_t* o2, const uint32_t* d2,
                            const DeviceBuffer<uint32_t>& caps,
                            GlobalDedupCsr& out, bool fetch_refs) {
      DeviceBuffer<uint64_t> ooff(owned + 1);
      uint64_t cap_total =
          exclusive_scan(caps.data(), ooff.data(), owned, s, I.scan);
      DeviceBuffer<uint32_t> oadj(cap_total ? cap_total : 1);
      DeviceBuffer<uint32_t> ocnt(owned ? owned : 1);
      ocnt.zero(s);
      if (n_small)
        lcc_orient_small_kernel<<<grid_for(n_small * kWave), kBlock, 0,
                                  s>>>(o1, d1, o2, d2, t_small.data(),
                                       n_small, g.v_begin, Dv.data(),
                                       ooff.data(), oadj.data(),
                                       ocnt.data(), false);
      if (n_mid)
        lcc_orient_mid_kernel<<<std::min<int>(n_mid, kMaxGrid), kBlock, 0,
                                s>>>(o1, d1, o2, d2, t_mid.data(), n_mid,
                                     g.v_begin, Dv.data(), ooff.data(),
                                     oadj.data(), ocnt.data(), false);
      if (n_wavelds)
        lcc_orient_wave_kernel<<<grid_for(((n_wavelds + 3) / 4) * kBlock),
                                 kBlock, 0, s>>>(
            o1, d1, o2, d2, t_wavelds.data(), n_wavelds, g.v_begin,
            Dv.data(), ooff.data(), oadj.data(), ocnt.data(), false);
      if (n_large) {
        fill(tbl_lab.data(), kCdlpEmpty, heavy_total, s);
        lcc_orient_large_kernel<<<std::min<int>(n_large, kMaxGrid), kBlock,
                                  0, s>>>(o1, d1, o2, d2, t_large.data(),
                                          n_large, heavy_off.data(),
                                          tbl_lab.data(), g.v_begin,
                                          Dv.data(), ooff.data(),
                                          oadj.data(), ocnt.data(), false);
      }
      build_global_csr(I, rank_, world_, nv_pad, slice, owned, g.v_begin,
                       ooff, ocnt, oadj, out, s,
                       fetch_refs ? &ridx : nullptr,
                       fetch_refs ? &roff : nullptr,
                       fetch_refs ? &sidx : nullptr,
                       fetch_refs ? &soff : nullptr);
    };

    // U family: in ∪ out, cap = D — never probed remotely, stays local
    {
      DeviceBuffer<uint32_t> caps(owned ? owned : 1);
      HIP_CHECK(hipMemcpyAsync(caps.data(), Dv.data() + g.v_begin,
                               owned * 4, hipMemcpyDeviceToDevice, s));
      build_family(off1, dst1, off2, dst2, caps, U, false);
    }
    // referenced u's = dsts of own U rows (the only rows whose O family
    // the count kernels probe)
    if (multi) {
      uint64_t ub = 0, ue = 0;
      HIP_CHECK(hipMemcpyAsync(&ub, U.goff.data() + g.v_begin, 8,
                               hipMemcpyDeviceToHost, s));
      HIP_CHECK(hipMemcpyAsync(&ue,
                               U.goff.data() + std::min<uint64_t>(
                                   g.v_end, nv_pad),
                               8, hipMemcpyDeviceToHost, s));
      HIP_CHECK(hipStreamSynchronize(s));
      size_t nwords = (static_cast<size_t>(nv_pad) + 31) / 32;
      DeviceBuffer<uint32_t> bm(nwords);
      bm.zero(s);
      if (ue > ub)
        mark_dsts_kernel<<<grid_for(ue - ub), kBlock, 0, s>>>(
            U.gdst.data() + ub, ue - ub, g.v_begin, g.v_end,
            DevBitmap{bm.data()});
      build_ref_lists(I, comm_, bm, nwords, slice, rank_, world_, ridx,
                      roff, sidx, soff, s);
    }
    // O family: distinct out, cap = min(out_deg, D)
    {
      DeviceBuffer<uint32_t> caps(owned ? owned : 1);
      if (owned)
        lcc_outcap_kernel<<<grid_for(owned), kBlock, 0, s>>>(
            off1, Dv.data(), owned, g.v_begin, caps.data());
      build_family(off1, dst1, nullptr, nullptr, caps, O, multi);
    }
  }

  DeviceBuffer<uint64_t> hoffU, hoffO;
  DeviceBuffer<uint32_t> htabU, htabO;
  build_hash_sets(I, nv_pad, U, hoffU, htabU, s);
  build_hash_sets(I, nv_pad, O, hoffO, htabO, s);

  DeviceBuffer<unsigned long long> Tcnt(nv_pad);
  Tcnt.zero(s);
  // min-side size above which an edge defers to the wave-per-edge pass;
  // 96 -> 48 after the two-box sweep + clean full-scale re-measurement
  // (profiles/r01_lcc_heavy_thresh_sweep.md, r02_lcc_heavy48_datagen90:
  // 2.62 s -> 2.50 s)
  uint32_t kHeavyThresh = 48;
  if (const char* ht = getenv("GRAPEHIP_LCC_HEAVY"))
    kHeavyThresh = static_cast<uint32_t>(atoi(ht));
  DeviceBuffer<unsigned long long> heavy_q;
  DeviceBuffer<unsigned long long> heavy_n(1);
  heavy_n.zero(s);
  {
    uint64_t lo = 0, hi = 0;
    HIP_CHECK(hipMemcpyAsync(&lo, U.goff.data() + g.v_begin, 8,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipMemcpyAsync(&hi,
                             U.goff.data() + std::min<uint64_t>(g.v_end,
                                                                nv_pad),
                             8, hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    heavy_q.resize(std::max<uint64_t>(hi - lo, 1));
  }
  if (owned) {
    DeviceBuffer<uint32_t> wee_rows(owned), big_rows(owned);
    DeviceBuffer<unsigned long long> tcnts(2);
    tcnts.zero(s);
    lcc_dir_bucket_kernel<<<grid_for(owned), kBlock, 0, s>>>(
        U.goff.data(), owned, g.v_begin, wee_rows.data(), tcnts.data() + 0,
        big_rows.data(), tcnts.data() + 1);
    auto tc = tcnts.download(s);
    if (tc[0])
      lcc_dir_count_wee_kernel<<<grid_for(tc[0] * kWave), kBlock, 0, s>>>(
          U.goff.data(), U.gdst.data(), hoffU.data(), htabU.data(),
          O.goff.data(), O.gdst.data(), hoffO.data(), htabO.data(),
          wee_rows.data(), tc[0], g.v_begin, Tcnt.data(), kHeavyThresh,
          heavy_q.data(), heavy_n.data());
    if (tc[1])
      lcc_dir_count_staged_kernel<<<std::min<uint64_t>(tc[1], kMaxGrid),
                                    kBlock, 0, s>>>(
          U.goff.data(), U.gdst.data(), hoffU.data(), htabU.data(),
          O.goff.data(), O.gdst.data(), hoffO.data(), htabO.data(),
          big_rows.data(), tc[1], g.v_begin, Tcnt.data(), kHeavyThresh,
          heavy_q.data(), heavy_n.data());
  }
  {
    unsigned long long hn = 0;
    HIP_CHECK(hipMemcpyAsync(&hn, heavy_n.data(), 8, hipMemcpyDeviceToHost,
                             s));
    HIP_CHECK(hipStreamSynchronize(s));
    if (hn)
      lcc_dir_heavy_kernel<<<grid_for(hn * kWave), kBlock, 0, s>>>(
          U.goff.data(), U.gdst.data(), hoffU.data(), htabU.data(),
          O.goff.data(), O.gdst.data(), hoffO.data(), htabO.data(),
          heavy_q.data(), hn, Tcnt.data());
  }
  DeviceBuffer<double> lcc_out(owned ? owned : 1);
  if (owned)
    lcc_dir_finalize_kernel<<<grid_for(owned), kBlock, 0, s>>>(
        Tcnt.data(), Dv.data(), owned, g.v_begin, lcc_out.data());
  HIP_CHECK(hipDeviceSynchronize());
  if (comm_) comm_->barrier();
  double t1 = wall_s();

  GpuRunResult res;
  res.bytes_p2p = I.dc.bytes_p2p - b_p2p0;
  res.bytes_coll = I.dc.bytes_coll - b_coll0;
  res.rounds = 1;
  res.seconds = comm_ ? comm_->allreduce_max_double(t1 - t0) : (t1 - t0);
  res.traversed_edges = g.input_edges;
  if (fetch) {
    res.f64.resize(owned);
    HIP_CHECK(hipMemcpyAsync(res.f64.data(), lcc_out.data(), owned * 8,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
  }
  return res;
}

GpuRunResult GpuContext::lcc(DeviceGraph& g, bool fetch) {
  RangeMarker _mk("grapehip::lcc");
  auto& I = *impl_;
  hipStream_t s = I.compute;
  uint32_t nv_pad = padded_nv(g, world_);
  uint32_t owned = g.owned();
  uint32_t slice = nv_pad / (world_ ? world_ : 1);
  bool multi = world_ > 1;
  if (g.directed) return lcc_directed(g, fetch);

  const uint64_t* off1 = g.oe_off.data();
  const uint32_t* dst1 = g.oe_dst.data();
  const uint64_t* off2 = nullptr;
  const uint32_t* dst2 = nullptr;

  // graph-sized working set (freed on return). Multi-GPU holds a PARTIAL
  // oriented CSR: own rows + the remote rows local oriented edges probe
  // (reference TransferAdjList staging, cuda/lcc/lcc.h:490-496, recast as
  // a one-shot referenced-row fetch) — memory and exchange volume scale
  // with the boundary, not with V.
  uint64_t oriented_total = 0;
  DeviceBuffer<uint32_t> Dv;                 // distinct degree, global idx
  DeviceBuffer<uint32_t> gcnt;               // oriented count (own+referenced)
  DeviceBuffer<uint64_t> goff;               // oriented CSR offsets (sparse)
  DeviceBuffer<uint32_t> gdst;               // oriented CSR dsts (partial)
  DeviceBuffer<unsigned long long> Tcnt;     // triangle credits
  // referenced-row request lists + staging for the count sync
  DeviceBuffer<uint32_t> ridx, sidx;
  std::vector<uint64_t> roff, soff;
  DeviceBuffer<uint8_t> rstage_s, rstage_r;
  if (multi) ensure_mirrors(I, comm_, g, rank_, world_, s);

  if (comm_) comm_->barrier();
  HIP_CHECK(hipDeviceSynchronize());
  double t0 = wall_s();
  const uint64_t b_p2p0 = I.dc.bytes_p2p, b_coll0 = I.dc.bytes_coll;

  // tier rows by combined degree (reuses the CDLP bucketer)
  DeviceBuffer<uint32_t> t_small(owned), t_mid(owned), t_large(owned);
  {
    DeviceBuffer<uint32_t> t_wavelds(owned ? owned : 1);
    DeviceBuffer<unsigned long long> cnts(5);
    cnts.zero(s);
    if (owned)
      cdlp_bucket_kernel<<<grid_for(owned), kBlock, 0, s>>>(
          off1, off2, owned, nullptr, cnts.data() + 3, t_small.data(),
          cnts.data() + 0, t_wavelds.data(), cnts.data() + 4, t_mid.data(),
          cnts.data() + 1, t_large.data(), cnts.data() + 2);
    auto hc = cnts.download(s);
    uint64_t n_small = hc[0], n_mid = hc[1], n_large = hc[2],
             n_wavelds = hc[4];

    // heavy-row hash pool (labels only)
    DeviceBuffer<uint32_t> heavy_caps;
    DeviceBuffer<uint64_t> heavy_off;
    DeviceBuffer<uint32_t> tbl_lab;
    uint64_t heavy_total = 0;
    if (n_large) {
      heavy_caps.resize(n_large);
      heavy_off.resize(n_large + 1);
      cdlp_heavy_cap_kernel<<<grid_for(n_large), kBlock, 0, s>>>(
          off1, off2, t_large.data(), n_large, heavy_caps.data());
      heavy_total = exclusive_scan(heavy_caps.data(), heavy_off.data(),
                                   n_large, s, I.scan);
      tbl_lab.resize(heavy_total);
    }

    // pass 1: distinct counts
    Dv.resize(nv_pad);
    Dv.zero(s);
    if (n_small)
      lcc_distinct_small_kernel<<<grid_for(n_small * kWave), kBlock, 0, s>>>(
          off1, dst1, off2, dst2, t_small.data(), n_small, g.v_begin,
          Dv.data());
    if (n_wavelds)
      lcc_distinct_wave_kernel<<<grid_for(((n_wavelds + 3) / 4) * kBlock),
                                 kBlock, 0, s>>>(off1, dst1, off2, dst2,
                                                 t_wavelds.data(),
                                                 n_wavelds, g.v_begin,
                                                 Dv.data());
    if (n_mid)
      lcc_distinct_mid_kernel<<<std::min<int>(n_mid, kMaxGrid), kBlock, 0,
                                s>>>(off1, dst1, off2, dst2, t_mid.data(),
                                     n_mid, g.v_begin, Dv.data());
    if (n_large) {
      fill(tbl_lab.data(), kCdlpEmpty, heavy_total, s);
      lcc_distinct_large_kernel<<<std::min<int>(n_large, kMaxGrid), kBlock, 0,
                                  s>>>(off1, dst1, off2, dst2, t_large.data(),
                                       n_large, heavy_off.data(),
                                       tbl_lab.data(), g.v_begin, Dv.data());
    }
    if (multi) {
      // the orient passes read Dv[dst] for local dsts only — refresh
      // referenced entries point-to-point instead of allgathering slices
      mirror_sync_begin(I, g, Dv.data(), s);
      mirror_sync_end(I, g, Dv.data(), s);
    }

    // pass 2: oriented adjacency in capacity layout (cap = D per row)
    DeviceBuffer<uint32_t> dcap(owned ? owned : 1);
    HIP_CHECK(hipMemcpyAsync(dcap.data(), Dv.data() + g.v_begin, owned * 4,
                             hipMemcpyDeviceToDevice, s));
    DeviceBuffer<uint64_t> ooff(owned + 1);
    uint64_t ocap_total =
        exclusive_scan(dcap.data(), ooff.data(), owned, s, I.scan);
    DeviceBuffer<uint32_t> oadj(ocap_total ? ocap_total : 1);
    DeviceBuffer<uint32_t> ocnt(owned ? owned : 1);
    ocnt.zero(s);
    if (n_small)
      lcc_orient_small_kernel<<<grid_for(n_small * kWave), kBlock, 0, s>>>(
          off1, dst1, off2, dst2, t_small.data(), n_small, g.v_begin,
          Dv.data(), ooff.data(), oadj.data(), ocnt.data(), true);
    if (n_mid)
      lcc_orient_mid_kernel<<<std::min<int>(n_mid, kMaxGrid), kBlock, 0, s>>>(
          off1, dst1, off2, dst2, t_mid.data(), n_mid, g.v_begin, Dv.data(),
          ooff.data(), oadj.data(), ocnt.data(), true);
    if (n_wavelds)
      lcc_orient_wave_kernel<<<grid_for(((n_wavelds + 3) / 4) * kBlock),
                               kBlock, 0, s>>>(
          off1, dst1, off2, dst2, t_wavelds.data(), n_wavelds, g.v_begin,
          Dv.data(), ooff.data(), oadj.data(), ocnt.data(), true);
    if (n_large) {
      fill(tbl_lab.data(), kCdlpEmpty, heavy_total, s);
      lcc_orient_large_kernel<<<std::min<int>(n_large, kMaxGrid), kBlock, 0,
                                s>>>(off1, dst1, off2, dst2, t_large.data(),
                                     n_large, heavy_off.data(),
                                     tbl_lab.data(), g.v_begin, Dv.data(),
                                     ooff.data(), oadj.data(), ocnt.data(),
                                     true);
    }

    // partial oriented CSR: own counts + counts of the rows local
    // oriented edges reference (synced over explicit request lists so
    // unfetched rows scan to zero length and the fetch regions stay
    // contiguous per owner)
    gcnt.resize(nv_pad);
    gcnt.zero(s);
    HIP_CHECK(hipMemcpyAsync(gcnt.data() + g.v_begin, ocnt.data(),
                             owned * 4, hipMemcpyDeviceToDevice, s));
    if (multi) {
      size_t nwords = (static_cast<size_t>(nv_pad) + 31) / 32;
      DeviceBuffer<uint32_t> bm(nwords);
      bm.zero(s);
      if (owned)
        lcc_mark_oriented_kernel<<<grid_for(static_cast<size_t>(owned) *
                                            kWave),
                                   kBlock, 0, s>>>(
            ooff.data(), ocnt.data(), oadj.data(), owned, g.v_begin,
            g.v_end, DevBitmap{bm.data()});
      uint64_t nref = build_ref_lists(I, comm_, bm, nwords, slice, rank_,
                                      world_, ridx, roff, sidx, soff, s);
      rstage_s.resize(soff[world_] * 4 + 4);
      rstage_r.resize(nref * 4 + 4);
      ref_sync_begin(I, sidx, soff, roff, rstage_s, rstage_r, gcnt.data(),
                     s);
      ref_sync_end(I, ridx, roff, rstage_r, gcnt.data(), s);
    }
    goff.resize(static_cast<size_t>(nv_pad) + 1);
    uint64_t g_total =
        exclusive_scan(gcnt.data(), goff.data(), nv_pad, s, I.scan);
    oriented_total = g_total;
    gdst.resize(g_total ? g_total : 1);
    if (owned)
      lcc_compact_kernel<<<grid_for(static_cast<size_t>(owned) * kWave),
                           kBlock, 0, s>>>(ooff.data(), ocnt.data(),
                                           oadj.data(), goff.data(), owned,
                                           g.v_begin, gdst.data());
    oadj.free();

    // per-row ascending sort
    DeviceBuffer<uint32_t> lds_rows(owned ? owned : 1),
        big_rows(owned ? owned : 1);
    DeviceBuffer<unsigned long long> scnt(2);
    scnt.zero(s);
    if (owned)
      lcc_sortbucket_kernel<<<grid_for(owned), kBlock, 0, s>>>(
          goff.data(), owned, g.v_begin, lds_rows.data(), scnt.data() + 0,
          big_rows.data(), scnt.data() + 1);
    auto hs = scnt.download(s);
    uint64_t n_lds = hs[0], n_big = hs[1];
    if (n_lds)
      lcc_sort_lds_kernel<<<std::min<int>(n_lds, kMaxGrid), kBlock, 0, s>>>(
          goff.data(), lds_rows.data(), n_lds, g.v_begin, gdst.data());
    if (n_big) {
      DeviceBuffer<uint32_t> caps(n_big);
      DeviceBuffer<uint64_t> pad_off(n_big + 1);
      lcc_bigcap_kernel<<<grid_for(n_big), kBlock, 0, s>>>(
          goff.data(), big_rows.data(), n_big, g.v_begin, caps.data());
      uint64_t pad_total =
          exclusive_scan(caps.data(), pad_off.data(), n_big, s, I.scan);
      DeviceBuffer<uint32_t> scratch(pad_total);
      lcc_bigpad_kernel<<<std::min<int>(n_big, kMaxGrid), kBlock, 0, s>>>(
          goff.data(), big_rows.data(), n_big, pad_off.data(), gdst.data(),
          g.v_begin, scratch.data());
      lcc_sort_big_kernel<<<std::min<int>(n_big, kMaxGrid), kBlock, 0, s>>>(
          pad_off.data(), n_big, scratch.data());
      lcc_bigunpad_kernel<<<std::min<int>(n_big, kMaxGrid), kBlock, 0, s>>>(
          goff.data(), big_rows.data(), n_big, pad_off.data(),
          scratch.data(), g.v_begin, gdst.data());
    }

    // fetch the referenced remote rows (sorted by their owners above):
    // each owner packs exactly the rows each peer requested; payloads
    // land directly in the requester's contiguous per-owner gdst region
    // (unfetched rows have zero length, so regions are hole-free)
    if (multi) {
      uint64_t ns = soff[world_];
      DeviceBuffer<uint32_t> slen(ns ? ns : 1);
      if (ns)
        gather4_kernel<<<grid_for(ns), kBlock, 0, s>>>(
            sidx.data(), ns, gcnt.data(), slen.data());
      DeviceBuffer<uint64_t> spack(ns + 1);
      uint64_t stotal =
          exclusive_scan(slen.data(), spack.data(), ns, s, I.scan);
      DeviceBuffer<uint32_t> srows(stotal ? stotal : 1);
      if (ns)
        lcc_pack_rows_kernel<<<grid_for(ns * kWave), kBlock, 0, s>>>(
            sidx.data(), ns, goff.data(), gdst.data(), spack.data(),
            srows.data());
      // per-peer byte offsets: send = packed scan at list boundaries,
      // recv = goff at slice boundaries (both derive from the same gcnt)
      std::vector<uint64_t> sb(world_ + 1), rb(world_ + 1);
      for (int f = 0; f <= world_; ++f) {
        uint64_t v = 0;
        HIP_CHECK(hipMemcpyAsync(&v, spack.data() + soff[f], 8,
                                 hipMemcpyDeviceToHost, s));
        HIP_CHECK(hipStreamSynchronize(s));
        sb[f] = v * 4;
        uint64_t idx = std::min<uint64_t>(
            static_cast<uint64_t>(f) * slice, nv_pad);
        HIP_CHECK(hipMemcpyAsync(&v, goff.data() + idx, 8,
                                 hipMemcpyDeviceToHost, s));
        HIP_CHECK(hipStreamSynchronize(s));
        rb[f] = v * 4;
      }
      I.dc.sendrecv(reinterpret_cast<const uint8_t*>(srows.data()), sb,
                    reinterpret_cast<uint8_t*>(gdst.data()), rb, s);
      HIP_CHECK(hipStreamSynchronize(s));
    }
  }

  // per-vertex oriented hash sets (intersection via O(1) probes)
  DeviceBuffer<uint32_t> hcap(nv_pad);
  DeviceBuffer<uint64_t> hoff(static_cast<size_t>(nv_pad) + 1);
  lcc_hashcap_kernel<<<grid_for(nv_pad), kBlock, 0, s>>>(goff.data(), nv_pad,
                                                         hcap.data());
  uint64_t htotal =
      exclusive_scan(hcap.data(), hoff.data(), nv_pad, s, I.scan);
  hcap.free();
  DeviceBuffer<uint32_t> htab(htotal);
  fill(htab.data(), kCdlpEmpty, htotal, s);
  lcc_hashfill_kernel<<<grid_for(static_cast<size_t>(nv_pad) * kWave),
                        kBlock, 0, s>>>(goff.data(), gdst.data(),
                                        hoff.data(), nv_pad, htab.data());

  // triangle counting: light edges inline, heavy edges deferred to a
  // wave-per-edge pass (power-law tail otherwise stalls single lanes)
  Tcnt.resize(nv_pad);
  Tcnt.zero(s);
  // min-side size above which an edge defers to the wave-per-edge pass;
  // 96 -> 48 after the two-box sweep + clean full-scale re-measurement
  // (profiles/r01_lcc_heavy_thresh_sweep.md, r02_lcc_heavy48_datagen90:
  // 2.62 s -> 2.50 s)
  uint32_t kHeavyThresh = 48;
  if (const char* ht = getenv("GRAPEHIP_LCC_HEAVY"))
    kHeavyThresh = static_cast<uint32_t>(atoi(ht));
  const bool nowit = getenv("GRAPEHIP_LCC_NOWIT") != nullptr;
  DeviceBuffer<unsigned long long> heavy_q;
  DeviceBuffer<unsigned long long> heavy_n(1);
  heavy_n.zero(s);
  {
    // worst case: every local oriented edge is heavy
    uint64_t local_oedges = 0;
    HIP_CHECK(hipMemcpyAsync(&local_oedges,
                             goff.data() + std::min<uint64_t>(
                                 static_cast<uint64_t>(g.v_end), nv_pad),
                             8, hipMemcpyDeviceToHost, s));
    uint64_t base_off = 0;
    HIP_CHECK(hipMemcpyAsync(&base_off, goff.data() + g.v_begin, 8,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
    heavy_q.resize(std::max<uint64_t>(local_oedges - base_off, 1));
  }
  if (owned) {
    DeviceBuffer<uint32_t> wee_rows(owned), big_rows(owned);
    DeviceBuffer<unsigned long long> tcnts(2);
    tcnts.zero(s);
    lcc_tribucket_kernel<<<grid_for(owned), kBlock, 0, s>>>(
        goff.data(), owned, g.v_begin, wee_rows.data(), tcnts.data() + 0,
        big_rows.data(), tcnts.data() + 1);
    auto tc = tcnts.download(s);
    if (tc[0])
      lcc_tri_wee_kernel<<<grid_for(tc[0] * kWave), kBlock, 0, s>>>(
          goff.data(), gdst.data(), hoff.data(), htab.data(),
          wee_rows.data(), tc[0], g.v_begin, Tcnt.data(), kHeavyThresh,
          heavy_q.data(), heavy_n.data(), nowit);
    if (tc[1])
      lcc_tri_staged_kernel<<<std::min<uint64_t>(tc[1], kMaxGrid), kBlock,
                              0, s>>>(
          goff.data(), gdst.data(), hoff.data(), htab.data(),
          big_rows.data(), tc[1], g.v_begin, Tcnt.data(), kHeavyThresh,
          heavy_q.data(), heavy_n.data(), nowit);
  }
  {
    unsigned long long hn = 0;
    HIP_CHECK(hipMemcpyAsync(&hn, heavy_n.data(), 8, hipMemcpyDeviceToHost,
                             s));
    HIP_CHECK(hipStreamSynchronize(s));
    if (getenv("GRAPEHIP_DEBUG"))
      fprintf(stderr, "[lcc] oriented_total=%llu heavy_edges=%llu\n",
              static_cast<unsigned long long>(oriented_total),
              static_cast<unsigned long long>(hn));
    if (hn)
      lcc_triangle_heavy_kernel<<<grid_for(hn * kWave), kBlock, 0, s>>>(
          goff.data(), gdst.data(), hoff.data(), htab.data(),
          heavy_q.data(), hn, Tcnt.data());
  }
  if (multi) I.dc.allreduce_sum_u64(Tcnt.data(), nv_pad, s);
  DeviceBuffer<double> lcc_out(owned ? owned : 1);
  if (owned)
    lcc_finalize_kernel<<<grid_for(owned), kBlock, 0, s>>>(
        Tcnt.data(), Dv.data(), owned, g.v_begin, lcc_out.data());
  HIP_CHECK(hipDeviceSynchronize());
  if (comm_) comm_->barrier();
  double t1 = wall_s();

  GpuRunResult res;
  res.bytes_p2p = I.dc.bytes_p2p - b_p2p0;
  res.bytes_coll = I.dc.bytes_coll - b_coll0;
  res.rounds = 1;
  res.seconds = comm_ ? comm_->allreduce_max_double(t1 - t0) : (t1 - t0);
  res.traversed_edges = g.input_edges;
  if (fetch) {
    res.f64.resize(owned);
    HIP_CHECK(hipMemcpyAsync(res.f64.data(), lcc_out.data(), owned * 8,
                             hipMemcpyDeviceToHost, s));
    HIP_CHECK(hipStreamSynchronize(s));
  }
  return res;
}

}  // namespace grapehip
