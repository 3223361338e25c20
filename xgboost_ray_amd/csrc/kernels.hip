// MI355X (gfx950, CDNA4) kernels for the gradient-boosted-tree engine.
//
// MI355X-native replacement for XGBoost's CUDA gpu_hist updater internals
// (the reference delegates them to libxgboost):
//   - quantize_gpair: fp32 gradient pairs -> int64 fixed point
//   - bin_matrix:     fp32 features -> uint8 quantile bins (LDS-cached cuts)
//   - gather + build_histogram: LDS-tiled per-(node,feature,bin) int64
//     gradient histograms, 64-wide-wave layout, deterministic by integer
//     associativity (no float atomics anywhere)
//   - find_splits:    sequential-per-(node,feature) scan, bitwise-identical
//     to the CPU torch reference (same add/divide order)
//   - partition_rows: stable two-pass partition (ballot prefix + scatter;
//     two-phase begin/finish entry, device-planned variant)
//   - grad_fused:     fused objective gradient + |g|/|h| block maxes
//   - predict_trees:  packed-node tree walk (LDS tree-tiled serving path)
//
// Design notes (per /opt/skills/guides/cdna_hip_programming.md):
//   wave = 64 lanes; LDS = 160 KiB/CU; histogram tiles sized so 1-2
//   workgroups fit per CU; all global traffic vectorized where layout
//   permits; grids sized >> 256 CUs.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <vector>

#define WAVE 64
#define CHECK_HIP(x)                                                         \
  do {                                                                       \
    hipError_t err = (x);                                                    \
    TORCH_CHECK(err == hipSuccess, "HIP error: ", hipGetErrorString(err));   \
  } while (0)

static inline int64_t ceil_div(int64_t a, int64_t b) { return (a + b - 1) / b; }

// torch spells the ROCm GPU device type with its legacy vendor name in the
// public tensor API; isolate that single spelling here.
static inline bool on_gpu(const torch::Tensor& t) { return t.is_cuda(); }

// ---------------------------------------------------------------------------
// quantize_gpair: [n,2] f32 -> [n,2] i64 (round to nearest)
// ---------------------------------------------------------------------------
// int32 pairs: |q| <= 2^30 by scale construction, so int32 holds each
// row exactly and int64 accumulators hold any sum - half the gradient
// traffic of int64 pairs at identical numerics.
__global__ void quantize_gpair_kernel(const float2* __restrict__ gpair,
                                      int2* __restrict__ out,
                                      double scale_g, double scale_h,
                                      int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float2 gp = gpair[i];
    int2 q;
    q.x = (int)llrint((double)gp.x * scale_g);
    q.y = (int)llrint((double)gp.y * scale_h);
    out[i] = q;
  }
}

// ---------------------------------------------------------------------------
// bin_matrix: values [n,F] -> bins u8 [n,F].
// bin = count of cuts <= v (upper_bound), clamped to nb-1; NaN -> 255.
// Cuts staged in LDS when they fit (F*max_bin floats; 200*255*4B = 204KB
// does NOT fit -> fall back to L2-resident global reads, still fast since
// cuts are tiny and re-read by every CU via L3).
// ---------------------------------------------------------------------------
template <bool LDS_CUTS>
__global__ void bin_matrix_kernel(const float* __restrict__ values,
                                  uint8_t* __restrict__ out,
                                  const float* __restrict__ cuts,
                                  const int64_t* __restrict__ cut_ptr,
                                  int64_t n, int F, int total_cuts,
                                  int64_t out_stride) {
  extern __shared__ float lds_cuts[];
  const float* C = cuts;
  if (LDS_CUTS) {
    for (int i = threadIdx.x; i < total_cuts; i += blockDim.x)
      lds_cuts[i] = cuts[i];
    __syncthreads();
    C = lds_cuts;
  }
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t total = n * (int64_t)F;
  for (; idx < total; idx += stride) {
    int f = (int)(idx % F);
    float v = values[idx];
    int lo = (int)cut_ptr[f], hi = (int)cut_ptr[f + 1];
    int nb = hi - lo;
    uint8_t b;
    if (isnan(v)) {
      b = 255;
    } else if (nb == 0) {
      b = 0;
    } else {
      // upper_bound: first cut > v
      int l = 0, r = nb;
      while (l < r) {
        int m = (l + r) >> 1;
        if (C[lo + m] <= v) l = m + 1; else r = m;
      }
      if (l > nb - 1) l = nb - 1;
      b = (uint8_t)l;
    }
    out[(idx / F) * out_stride + f] = b;
  }
}

// ---------------------------------------------------------------------------
// gather_gpair_seg: gpair_seg[i] = gpair_q[ridx[i]] for i in [0, n_seg)
// (coalesced writes; the gather read hits L2/L3). Done once per depth so
// the histogram kernel reads gradients coalesced per feature block.
// ---------------------------------------------------------------------------
__global__ void gather_gpair_kernel(const int2* __restrict__ gpair,
                                    const int32_t* __restrict__ ridx,
                                    int2* __restrict__ out, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) out[i] = gpair[(int64_t)(uint32_t)ridx[i]];
}

// ---------------------------------------------------------------------------
// build_histogram: LDS-tiled int64 gradient-pair histograms.
//
// Grid: x = flattened (node, row-chunk), y = feature block.
// Each workgroup accumulates a [FB][n_bins][2] int64 tile in LDS with
// ds-atomics, then merges once into the global histogram with device
// atomics. Integer accumulation => order-independent, bitwise
// deterministic (the checkpoint-determinism contract).
// ---------------------------------------------------------------------------
#define HIST_THREADS 512
static int hist_rows_per_wg() {
  static int v = [] {
    const char* e = getenv("RXGB_HIST_ROWS");
    int r = e ? atoi(e) : 16384;
    return (r >= 1024 && r <= 262144) ? r : 16384;
  }();
  return v;
}
#define HIST_ROWS_PER_WG hist_rows_per_wg()

// VEC16: the binned matrix row stride is 16-byte aligned (padded layout,
// pad bytes == 255) -> one uint4 load fetches a whole 16-feature block of
// a row. Lane rotation spreads one wave-instruction's LDS atomics over 16
// different features (max 4 lanes per feature histogram) so same-bin
// serialization on skewed features drops ~16x.
// One row's 16-feature uint4 -> LDS tile atomics with word-granular lane
// rotation (shared by the single-block and multi-block kernels below).
//
// LDS layout is SoA: g-plane at [f*n_bins + b], h-plane at
// [16*n_bins + f*n_bins + b]. Interleaved (g,h) pairs put the bank index
// at (b*4)%64 -> only bin%16 distinguishes banks; the split doubles the
// spread to bin%32 (PMC: conflict/active 0.74 interleaved).
__device__ __forceinline__ void hist_accum_packed16(
    unsigned long long* lds_hist, uint4 packed, int n_bins, int lane,
    longlong2 gp) {
  const int hplane = 16 * n_bins;
  const int r4 = lane & 3;
  const uint32_t w0 = packed.x, w1 = packed.y, w2 = packed.z, w3 = packed.w;
  const bool s1 = (r4 & 1) != 0, s2 = (r4 & 2) != 0;
  const uint32_t t01 = s1 ? w1 : w0, t23 = s1 ? w3 : w2;
  const uint32_t u01 = s1 ? w2 : w1, u23 = s1 ? w0 : w3;
  const uint32_t rw0 = s2 ? t23 : t01;  // w[(0+r4)&3]
  const uint32_t rw1 = s2 ? u23 : u01;  // w[(1+r4)&3]
  const uint32_t rw2 = s2 ? t01 : t23;  // w[(2+r4)&3]
  const uint32_t rw3 = s2 ? u01 : u23;  // w[(3+r4)&3]
  const uint32_t rws[4] = {rw0, rw1, rw2, rw3};
  // byte-granular second rotation: the 16 lanes sharing a word-rotation
  // phase split 4 ways over the word's features (one v_alignbit), so one
  // wave instruction hits 16 distinct features instead of 4 -> 4x fewer
  // same-bin collisions. The dynamic rotate keeps everything in named
  // registers (rule 20: no runtime-indexed arrays).
  const uint32_t rsh = 8u * ((lane >> 2) & 3);
  const int r2 = (lane >> 2) & 3;
  #pragma unroll
  for (int jj = 0; jj < 4; ++jj) {
    const uint32_t w = rws[jj];
    const uint32_t wr = (w >> rsh) | (w << ((32u - rsh) & 31u));
    const int fw = ((jj + r4) & 3) * 4;
    #pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      const int b = (wr >> (8 * kk)) & 0xFF;
      if (b != 255) {
        const int f = fw + ((kk + r2) & 3);
        // XOR swizzle: bank index (cell*2)%64 was feature-independent;
        // ^ (f&7) staggers features across 8 bank offsets. Only valid
        // when b^7 cannot leave the feature's bin range (n_bins==256).
        const int cell =
            (f * n_bins + b) ^ (n_bins == 256 ? (f & 7) : 0);
        unsigned long long* c = &lds_hist[cell];
        atomicAdd(c, (unsigned long long)gp.x);
        atomicAdd(c + hplane, (unsigned long long)gp.y);
      }
    }
  }
}

__device__ __forceinline__ void hist_accum_row16(
    unsigned long long* lds_hist, const uint8_t* __restrict__ bins,
    uint64_t r, int64_t row_stride, int f0, int n_bins, int lane,
    longlong2 gp) {
  const uint4 packed =
      *reinterpret_cast<const uint4*>(bins + r * row_stride + f0);
  hist_accum_packed16(lds_hist, packed, n_bins, lane, gp);
}

// Tiny-node fast path: accumulate a node's rows straight into the
// GLOBAL histogram. For a node with c rows the LDS route pays a
// 16*n_bins*2-cell tile zero + scan per workgroup regardless of c; for
// c*2*16 atomics < that cost (c below ~512 at 256 bins) direct global
// int64 atomics are cheaper - and identically deterministic (integer
// adds in any order). The caller zeroes hist, so partial adds compose.
__device__ inline void hist_direct_rows(
    const uint8_t* __restrict__ bins, const int2* __restrict__ gpair_seg,
    const int32_t* __restrict__ ridx, long long* __restrict__ ghist_node,
    int64_t seg_start, int64_t row_lo, int64_t row_hi, int64_t row_stride,
    int f0, int fcount, int n_bins, int tid, int nthreads) {
  for (int64_t i = row_lo + tid; i < row_hi; i += nthreads) {
    const int64_t seg_i = seg_start + i;
    const int2 gpi = gpair_seg[seg_i];
    const uint64_t r = (uint32_t)ridx[seg_i];
    const uint8_t* rowb = bins + r * row_stride + f0;
    #pragma unroll 4
    for (int f = 0; f < fcount; ++f) {
      const int b = rowb[f];
      if (b != 255) {
        long long* cell =
            &ghist_node[(((size_t)(f0 + f)) * n_bins + b) * 2];
        atomicAdd((unsigned long long*)cell,
                  (unsigned long long)(long long)gpi.x);
        atomicAdd((unsigned long long*)cell + 1,
                  (unsigned long long)(long long)gpi.y);
      }
    }
  }
}

template <int VFB>  // 16: uint4 row loads; 8: uint2; 0: byte fallback
__global__ __launch_bounds__(HIST_THREADS) void build_histogram_kernel(
    const uint8_t* __restrict__ bins,        // [n_rows_total, row_stride]
    const int2* __restrict__ gpair_seg,      // [seg_total] segment order
    const int32_t* __restrict__ ridx,        // [seg_total]
    const int64_t* __restrict__ node_start,  // [K] segment starts
    const int64_t* __restrict__ chunk_off,   // [K+1] cumulative chunks
    long long* __restrict__ hist,            // [K, F, n_bins, 2]
    int K, int F, int n_bins, int fb_size, int64_t row_stride, int f_base,
    int rows_per_wg, int direct_rows) {
  // locate (node, chunk) from blockIdx.x via binary search on chunk_off
  int wg = blockIdx.x;
  int lo = 0, hi = K;
  while (lo + 1 < hi) {
    int m = (lo + hi) >> 1;
    if (chunk_off[m] <= wg) lo = m; else hi = m;
  }
  const int node = lo;
  const int64_t chunk_in_node = wg - chunk_off[node];
  const int64_t seg_start = node_start[node];
  // rows of this chunk within the node's segment
  const int64_t row_lo = chunk_in_node * (int64_t)rows_per_wg;

  const int fb = blockIdx.y;
  const int f0 = f_base + fb * fb_size;
  const int fcount = fb_size < (F - f0) ? fb_size : (F - f0);

  {
    const int64_t node_count_e = node_start[K + node];
    if (node_count_e < (int64_t)direct_rows) {
      int64_t row_hi_e = row_lo + rows_per_wg;
      if (row_hi_e > node_count_e) row_hi_e = node_count_e;
      hist_direct_rows(bins, gpair_seg, ridx,
                       hist + ((size_t)node * F) * (size_t)n_bins * 2,
                       seg_start, row_lo, row_hi_e, row_stride, f0, fcount,
                       n_bins, threadIdx.x, blockDim.x);
      return;
    }
  }

  extern __shared__ unsigned long long lds_hist[];  // [fb_size][n_bins][2]
  // VFB==16 uses the SoA g/h-plane layout (see hist_accum_row16): zero
  // both full planes even for a partial last block.
  const int tile = (VFB == 16 ? 16 : fcount) * n_bins * 2;
  for (int i = threadIdx.x; i < tile; i += blockDim.x) lds_hist[i] = 0ull;
  __syncthreads();

  // row loop: lanes take consecutive rows -> gpair reads coalesced
  // (segment order), bin reads are per-row contiguous byte runs.
  // counts are stored at node_start[K..2K) (cat_start_count layout)
  const int64_t node_count = node_start[K + node];
  int64_t row_hi = row_lo + rows_per_wg;
  if (row_hi > node_count) row_hi = node_count;

  const int lane = threadIdx.x & (WAVE - 1);
  for (int64_t i = row_lo + threadIdx.x; i < row_hi; i += blockDim.x) {
    const int64_t seg_i = seg_start + i;
    const int2 gpi = gpair_seg[seg_i];
    const longlong2 gp = {(long long)gpi.x, (long long)gpi.y};
    const uint64_t r = (uint32_t)ridx[seg_i];
    if constexpr (VFB == 16) {
      // fb_size == 16 and row base 16B-aligned by construction.
      // Word-granular lane rotation: lane l processes its row's feature
      // words in order (l&3), (l&3)+1, ... so one wave-instruction's LDS
      // atomics spread over 4 feature groups (4x fewer same-address
      // serializations on skewed features). The rotation is done with
      // branchless selects on NAMED registers - a runtime-indexed byte
      // array would go to scratch (5x slowdown).
      hist_accum_row16(lds_hist, bins, r, row_stride, f0, n_bins, lane, gp);
    } else if constexpr (VFB == 8) {
      const uint2 packed =
          *reinterpret_cast<const uint2*>(bins + r * row_stride + f0);
      const int r2 = lane & 1;
      const uint32_t w0 = packed.x, w1 = packed.y;
      const uint32_t rw0 = r2 ? w1 : w0;  // w[(0+r2)&1]
      const uint32_t rw1 = r2 ? w0 : w1;  // w[(1+r2)&1]
      const uint32_t rws[2] = {rw0, rw1};
      #pragma unroll
      for (int jj = 0; jj < 2; ++jj) {
        const uint32_t w = rws[jj];
        const int fw = ((jj + r2) & 1) * 4;
        #pragma unroll
        for (int kk = 0; kk < 4; ++kk) {
          const int b = (w >> (8 * kk)) & 0xFF;
          if (b != 255) {
            const int f = fw + kk;
            unsigned long long* cell =
                &lds_hist[((size_t)f * n_bins + b) * 2];
            atomicAdd(cell, (unsigned long long)gp.x);
            atomicAdd(cell + 1, (unsigned long long)gp.y);
          }
        }
      }
    } else {
      const uint8_t* rowb = bins + r * row_stride + f0;
      #pragma unroll 4
      for (int f = 0; f < fcount; ++f) {
        const int b = rowb[f];
        if (b != 255) {
          unsigned long long* cell = &lds_hist[((size_t)f * n_bins + b) * 2];
          atomicAdd(cell, (unsigned long long)gp.x);
          atomicAdd(cell + 1, (unsigned long long)gp.y);
        }
      }
    }
  }
  __syncthreads();

  // merge LDS tile into the global (interleaved [.,2]) histogram
  long long* ghist =
      hist + (((size_t)node * F + f0) * n_bins) * 2;
  if constexpr (VFB == 16) {
    // re-interleave while merging: consecutive lanes write consecutive
    // global u64s (a strided 2i/2i+1 pattern doubled global transactions)
    const int n2 = fcount * n_bins * 2;
    const int hplane = 16 * n_bins;
    for (int j = threadIdx.x; j < n2; j += blockDim.x) {
      const int cell = j >> 1;
      const int scell =
          cell ^ (n_bins == 256 ? ((cell / n_bins) & 7) : 0);
      const unsigned long long v = lds_hist[(j & 1) * hplane + scell];
      if (v) atomicAdd((unsigned long long*)&ghist[j], v);
    }
  } else {
    for (int i = threadIdx.x; i < tile; i += blockDim.x) {
      const unsigned long long v = lds_hist[i];
      if (v != 0ull) {
        atomicAdd((unsigned long long*)&ghist[i], v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// build_histogram_multifb: multi-feature-block variant for wide matrices.
//
// At F=200 (13 blocks of 16) the baseline kernel re-reads gpair_seg (8 B)
// + ridx (4 B) per row per block: 156 B/row/depth where only 12 B are
// needed. Here ONE workgroup loops over every feature block of its range,
// keeping its R rows/thread of gradient pairs + row indices in registers
// across blocks. Occupancy is LDS-bound anyway (64 KiB tile -> 2 WGs/CU
// = 4 waves/SIMD), so up to ~128 VGPRs the caching is free.
// Same LDS tile, same lane rotation, same int64 atomics => bitwise
// identical histograms to the baseline kernel.
// ---------------------------------------------------------------------------
template <int R>  // rows cached per thread; rows_per_wg = R * HIST_THREADS
__global__ __launch_bounds__(HIST_THREADS) void build_histogram_multifb_kernel(
    const uint8_t* __restrict__ bins,        // [n_rows_total, row_stride]
    const int2* __restrict__ gpair_seg,      // [seg_total] segment order
    const int32_t* __restrict__ ridx,        // [seg_total]
    const int64_t* __restrict__ node_start,  // [K] starts, [K..2K) counts
    const int64_t* __restrict__ chunk_off,   // [K+1] cumulative chunks
    long long* __restrict__ hist,            // [K, F, n_bins, 2]
    int K, int F, int n_bins, int n_fb, int64_t row_stride, int f_base) {
  int wg = blockIdx.x;
  int lo = 0, hi = K;
  while (lo + 1 < hi) {
    int m = (lo + hi) >> 1;
    if (chunk_off[m] <= wg) lo = m; else hi = m;
  }
  const int node = lo;
  const int64_t chunk_in_node = wg - chunk_off[node];
  const int64_t seg_start = node_start[node];
  const int64_t row_lo = chunk_in_node * (int64_t)(R * HIST_THREADS);
  const int64_t node_count = node_start[K + node];
  int64_t row_hi = row_lo + (int64_t)(R * HIST_THREADS);
  if (row_hi > node_count) row_hi = node_count;

  // Cache this thread's rows once. Strided assignment (i = row_lo + tid +
  // rr*512) keeps the loads coalesced per wave; validity is a prefix in
  // rr, so the block loop can predicate on rr < nvalid.
  int2 gp_reg[R];
  uint32_t rx_reg[R];
  int nvalid = 0;
  #pragma unroll
  for (int rr = 0; rr < R; ++rr) {
    const int64_t i = row_lo + threadIdx.x + (int64_t)rr * HIST_THREADS;
    if (i < row_hi) {
      const int64_t s = seg_start + i;
      gp_reg[rr] = gpair_seg[s];
      rx_reg[rr] = (uint32_t)ridx[s];
      nvalid = rr + 1;
    }
  }

  extern __shared__ unsigned long long lds_hist[];
  const int tile_full = 16 * n_bins * 2;
  const int hplane = 16 * n_bins;
  const int lane = threadIdx.x & (WAVE - 1);
  for (int blk = 0; blk < n_fb; ++blk) {
    const int f0 = f_base + blk * 16;
    for (int i = threadIdx.x; i < tile_full; i += blockDim.x)
      lds_hist[i] = 0ull;
    __syncthreads();
    #pragma unroll
    for (int rr = 0; rr < R; ++rr) {
      if (rr < nvalid) {
        const longlong2 gp = {(long long)gp_reg[rr].x,
                              (long long)gp_reg[rr].y};
        hist_accum_row16(lds_hist, bins, (uint64_t)rx_reg[rr], row_stride,
                         f0, n_bins, lane, gp);
      }
    }
    __syncthreads();
    // merge only real features (partial last block: pad bins==255 were
    // skipped in accumulation, but hist rows past F-1 must not be touched);
    // LDS is SoA g/h planes, global stays interleaved [.,2]
    const int fcount = 16 < (F - f0) ? 16 : (F - f0);
    const int n2 = fcount * n_bins * 2;
    long long* ghist = hist + (((size_t)node * F + f0) * n_bins) * 2;
    for (int j = threadIdx.x; j < n2; j += blockDim.x) {
      const int cell = j >> 1;
      const int scell =
          cell ^ (n_bins == 256 ? ((cell / n_bins) & 7) : 0);
      const unsigned long long v = lds_hist[(j & 1) * hplane + scell];
      if (v) atomicAdd((unsigned long long*)&ghist[j], v);
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// build_histogram_xcd: XCD-swizzled (chunk, block) grid for wide matrices.
//
// The multifb kernel re-fetches each row's bin lines once per feature
// block over a long window - no L2 survival. Here each (row-chunk,
// feature-block) pair is its own workgroup again, but the flat index is
// swizzled so ALL n_fb blocks of one chunk land on the SAME XCD
// (consecutive blockIdx round-robin over the 8 XCDs; stride-8 indices
// share one): the chunk's bin lines are fetched once into that XCD's L2
// and every 16 B uint4 of each 64 B line is consumed by a co-resident
// workgroup. gpair_seg/ridx re-reads hit L2 the same way (the 13 blocks
// read the same 48 KB of segment data).
// ---------------------------------------------------------------------------
template <int THREADS>
__global__ __launch_bounds__(THREADS) void build_histogram_xcd_kernel(
    const uint8_t* __restrict__ bins,        // [n_rows_total, row_stride]
    const int2* __restrict__ gpair_seg,      // [seg_total] segment order
    const int32_t* __restrict__ ridx,        // [seg_total]
    const int64_t* __restrict__ node_start,  // [K] starts, [K..2K) counts
    const int64_t* __restrict__ chunk_off,   // [K+1] cumulative chunks
    long long* __restrict__ hist,            // [K, F, n_bins, 2]
    int K, int F, int n_bins, int n_fb, int64_t row_stride, int f_base,
    int rows_per_wg, int total_chunks, int direct_rows) {
  const int flat = blockIdx.x;
  const int octet = flat / (8 * n_fb);
  const int rem = flat % (8 * n_fb);
  const int fb = rem / 8;
  const int wg = octet * 8 + (rem % 8);  // row-chunk id
  if (wg >= total_chunks) return;        // tail of the last chunk-octet

  int lo = 0, hi = K;
  while (lo + 1 < hi) {
    int m = (lo + hi) >> 1;
    if (chunk_off[m] <= wg) lo = m; else hi = m;
  }
  const int node = lo;
  const int64_t chunk_in_node = wg - chunk_off[node];
  const int64_t seg_start = node_start[node];
  const int64_t row_lo = chunk_in_node * (int64_t)rows_per_wg;
  const int f0 = f_base + fb * 16;
  const int fcount = 16 < (F - f0) ? 16 : (F - f0);

  {
    const int64_t node_count_e = node_start[K + node];
    if (node_count_e < (int64_t)direct_rows) {
      int64_t row_hi_e = row_lo + rows_per_wg;
      if (row_hi_e > node_count_e) row_hi_e = node_count_e;
      hist_direct_rows(bins, gpair_seg, ridx,
                       hist + ((size_t)node * F) * (size_t)n_bins * 2,
                       seg_start, row_lo, row_hi_e, row_stride, f0, fcount,
                       n_bins, threadIdx.x, blockDim.x);
      return;
    }
  }

  extern __shared__ unsigned long long lds_hist[];
  const int hplane = 16 * n_bins;
  for (int i = threadIdx.x; i < 2 * hplane; i += blockDim.x)
    lds_hist[i] = 0ull;
  __syncthreads();

  const int64_t node_count = node_start[K + node];
  int64_t row_hi = row_lo + rows_per_wg;
  if (row_hi > node_count) row_hi = node_count;
  const int lane = threadIdx.x & (WAVE - 1);
  // 2-row software pipeline: both rows' random uint4 gathers are in
  // flight before either LDS-atomic burst starts (SQ_WAIT_ANY was 17x
  // SQ_BUSY with one outstanding gather per thread)
  for (int64_t i = row_lo + threadIdx.x; i < row_hi;
       i += 2 * (int64_t)blockDim.x) {
    const int64_t i2 = i + blockDim.x;
    const int64_t seg_i = seg_start + i;
    const int2 gpi = gpair_seg[seg_i];
    const uint64_t r1 = (uint32_t)ridx[seg_i];
    const uint4 p1 = *reinterpret_cast<const uint4*>(
        bins + r1 * row_stride + f0);
    const bool v2 = i2 < row_hi;
    int2 gpi2 = {0, 0};
    uint4 p2 = {0, 0, 0, 0};
    if (v2) {
      const int64_t seg_i2 = seg_start + i2;
      gpi2 = gpair_seg[seg_i2];
      const uint64_t r2 = (uint32_t)ridx[seg_i2];
      p2 = *reinterpret_cast<const uint4*>(bins + r2 * row_stride + f0);
    }
    hist_accum_packed16(lds_hist, p1, n_bins, lane,
                        {(long long)gpi.x, (long long)gpi.y});
    if (v2)
      hist_accum_packed16(lds_hist, p2, n_bins, lane,
                          {(long long)gpi2.x, (long long)gpi2.y});
  }
  __syncthreads();

  const int n2 = fcount * n_bins * 2;
  long long* ghist = hist + (((size_t)node * F + f0) * n_bins) * 2;
  for (int j = threadIdx.x; j < n2; j += blockDim.x) {
    const int cell = j >> 1;
    const int scell =
        cell ^ (n_bins == 256 ? ((cell / n_bins) & 7) : 0);
    const unsigned long long v = lds_hist[(j & 1) * hplane + scell];
    if (v) atomicAdd((unsigned long long*)&ghist[j], v);
  }
}

// ---------------------------------------------------------------------------
// find_splits: one thread per (node, feature); sequential 256-bin scan in
// the exact same FP order as the CPU torch reference (int64 -> double per
// bin, then left-to-right adds) so CPU and GPU grow identical trees.
// ---------------------------------------------------------------------------
struct SplitCand {
  double gain;
  int bin;
  int default_left;
  long long left_g;
  long long left_h;
};

__device__ inline double calc_score(double G, double H, double lam, double alpha) {
  if (alpha > 0.0) {
    double t = fabs(G) - alpha;
    G = t > 0.0 ? copysign(t, G) : 0.0;
  }
  double denom = H + lam;
  return denom > 0.0 ? G * G / denom : 0.0;
}

// One WAVE per (node, feature): 64 lanes cooperatively stage the 256-bin
// histogram into LDS (coalesced), then lane 0 runs the sequential scan
// from LDS in the exact CPU FP order. ~100x the parallelism of a
// thread-per-(k,f) scan at shallow depths, with identical numerics.
__device__ inline double calc_weight_d(double G, double H, double lam,
                                       double alpha) {
  if (alpha > 0.0) {
    double t = fabs(G) - alpha;
    G = t > 0.0 ? copysign(t, G) : 0.0;
  }
  double denom = H + lam;
  return denom > 0.0 ? -G / denom : 0.0;
}

__global__ __launch_bounds__(256) void find_splits_kf_kernel(
    const long long* __restrict__ hist,  // [K, F, B, 2]
    const long long* __restrict__ parent_g, const long long* __restrict__ parent_h,
    const int32_t* __restrict__ feat_bins, double scale_g, double scale_h,
    double lam, double alpha, double gamma, double mcw,
    const int8_t* __restrict__ mono,    // [F] or nullptr
    const double* __restrict__ bounds,  // [K, 2] node weight bounds
    const uint8_t* __restrict__ allowed,  // [K, F] interaction gate or null
    double* __restrict__ out_gain,     // [K, F]
    int32_t* __restrict__ out_bin,     // [K, F]
    uint8_t* __restrict__ out_dl,      // [K, F]
    long long* __restrict__ out_lg,    // [K, F]
    long long* __restrict__ out_lh,    // [K, F]
    int K, int F, int B) {
  // ONE 64-lane wave per (node, feature). The per-bin left sums come
  // from an INT64 inclusive wave-scan (lane-local serial prefix over a
  // contiguous bin chunk + shfl_up scan of lane totals) - integer adds
  // are exact under any order, so the scan reassociation cannot change a
  // single bit. Each bin's gain is then an independent f64 expression of
  // its exact prefix (mirrored 1:1 by the CPU oracle, which dequantizes
  // the same int cumsum), so 64 lanes evaluate 4 bins each in parallel
  // where the previous design ran a 255-step dependent f64 chain on 2
  // lanes (52% issue-stall, and nearly histogram-sized total time).
  const int wave = threadIdx.x / WAVE;
  const int lane = threadIdx.x % WAVE;
  const int waves_per_block = blockDim.x / WAVE;
  int64_t kf = (int64_t)blockIdx.x * waves_per_block + wave;
  if (kf >= (int64_t)K * F) return;
  const int k = (int)(kf / F);
  const int f = (int)(kf % F);
  if (allowed != nullptr && allowed[kf] == 0) {
    // interaction constraints: disallowed (node, feature) -> the same
    // no-split sentinel the CPU oracle produces (gain -1, dl 0)
    if (lane == 0) {
      out_gain[kf] = -1.0;
      out_bin[kf] = 0;
      out_dl[kf] = 0;
      out_lg[kf] = 0;
      out_lh[kf] = 0;
    }
    return;
  }
  const double inv_g = 1.0 / scale_g;
  const double inv_h = 1.0 / scale_h;
  const long long* gh = hist + ((size_t)k * F + f) * B * 2;
  const int nb = feat_bins[f];

  // lane-local inclusive prefixes over this lane's contiguous bin chunk
  const int bpl = (B + WAVE - 1) / WAVE;  // <= 4 for B <= 256
  long long lpg[4], lph[4];
  long long gsum = 0, hsum = 0;
  #pragma unroll
  for (int j = 0; j < 4; ++j) {
    if (j < bpl) {
      const int b = lane * bpl + j;
      if (b < B) {
        gsum += gh[b * 2];
        hsum += gh[b * 2 + 1];
      }
      lpg[j] = gsum;
      lph[j] = hsum;
    }
  }
  // inclusive wave-scan of lane totals -> exclusive offset per lane
  long long gscan = gsum, hscan = hsum;
  #pragma unroll
  for (int d = 1; d < WAVE; d <<= 1) {
    const long long ug = __shfl_up(gscan, d, WAVE);
    const long long uh = __shfl_up(hscan, d, WAVE);
    if (lane >= d) {
      gscan += ug;
      hscan += uh;
    }
  }
  const long long goff = gscan - gsum;   // exclusive prefix offset
  const long long hoff = hscan - hsum;
  const long long Gtot_q = __shfl(gscan, WAVE - 1, WAVE);
  const long long Htot_q = __shfl(hscan, WAVE - 1, WAVE);

  const double Gp = (double)parent_g[k] * inv_g;
  const double Hp = (double)parent_h[k] * inv_h;
  const double parent_score = calc_score(Gp, Hp, lam, alpha);
  const double Gmiss = Gp - (double)Gtot_q * inv_g;
  const double Hmiss = Hp - (double)Htot_q * inv_h;
  const long long Gmiss_q = parent_g[k] - Gtot_q;
  const long long Hmiss_q = parent_h[k] - Htot_q;

  const int c = (mono != nullptr) ? mono[f] : 0;
  double blo = 0.0, bup = 0.0;
  if (c != 0) {
    blo = bounds[(size_t)k * 2];
    bup = bounds[(size_t)k * 2 + 1];
  }

  // candidate ordering = CPU oracle order: gain desc, then dl=1 before
  // dl=0 (strictly-greater replacement across the two passes), then
  // lowest bin (ascending scan, strictly-greater replacement)
  SplitCand best = {-1.0, 0x7fffffff, -1, 0, 0};
  #pragma unroll
  for (int j = 0; j < 4; ++j) {
    if (j >= bpl) break;
    const int b = lane * bpl + j;
    if (b >= nb - 1 || b >= B) continue;
    const long long GLq = goff + lpg[j];
    const long long HLq = hoff + lph[j];
    const double GLd = (double)GLq * inv_g;
    const double HLd = (double)HLq * inv_h;
    #pragma unroll
    for (int dl = 1; dl >= 0; --dl) {
      const double gl = dl ? GLd + Gmiss : GLd;
      const double hl = dl ? HLd + Hmiss : HLd;
      const double gr = Gp - gl, hr = Hp - hl;
      if (hl < mcw || hr < mcw) continue;
      if (c != 0) {
        double wl = calc_weight_d(gl, hl, lam, alpha);
        double wr = calc_weight_d(gr, hr, lam, alpha);
        wl = fmin(fmax(wl, blo), bup);
        wr = fmin(fmax(wr, blo), bup);
        if (c > 0 ? (wl > wr) : (wl < wr)) continue;
      }
      const double gain = 0.5 * (calc_score(gl, hl, lam, alpha) +
                                 calc_score(gr, hr, lam, alpha) -
                                 parent_score) -
                          gamma;
      const bool take =
          (gain > best.gain) ||
          (gain == best.gain &&
           (dl > best.default_left ||
            (dl == best.default_left && b < best.bin)));
      if (take) {
        best.gain = gain;
        best.bin = b;
        best.default_left = dl;
        best.left_g = dl ? GLq + Gmiss_q : GLq;
        best.left_h = dl ? HLq + Hmiss_q : HLq;
      }
    }
  }

  // wave argmax reduction with the same comparator
  #pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    const double og = __shfl_down(best.gain, off, WAVE);
    const int ob = __shfl_down(best.bin, off, WAVE);
    const int odl = __shfl_down(best.default_left, off, WAVE);
    const long long olg = __shfl_down(best.left_g, off, WAVE);
    const long long olh = __shfl_down(best.left_h, off, WAVE);
    const bool take =
        (og > best.gain) ||
        (og == best.gain &&
         (odl > best.default_left ||
          (odl == best.default_left && ob < best.bin)));
    if (take) {
      best.gain = og;
      best.bin = ob;
      best.default_left = odl;
      best.left_g = olg;
      best.left_h = olh;
    }
  }
  if (lane != 0) return;
  if (best.gain <= -1.0) {
    // no-split sentinel keeps the CPU contract (gain -1, dl 0, bin 0)
    best.bin = 0;
    best.default_left = 0;
    best.left_g = 0;
    best.left_h = 0;
  }
  out_gain[kf] = best.gain;
  out_bin[kf] = best.bin;
  out_dl[kf] = (uint8_t)best.default_left;
  out_lg[kf] = best.left_g;
  out_lh[kf] = best.left_h;
}

// packs the per-node best split into ONE int64 [K, 6] row:
// (f32 gain bits, feature, bin, default_left, left_g, left_h) so the
// driver retrieves the whole depth's splits with a single D2H copy
// (dozens of tiny pageable copies per depth were the training loop's
// dominant host-side cost).
__global__ void find_splits_reduce_kernel(
    const double* __restrict__ kf_gain, const int32_t* __restrict__ kf_bin,
    const uint8_t* __restrict__ kf_dl, const long long* __restrict__ kf_lg,
    const long long* __restrict__ kf_lh,
    long long* __restrict__ out_packed,  // [K, 6]
    int K, int F) {
  int k = blockIdx.x * blockDim.x + threadIdx.x;
  if (k >= K) return;
  double bg = -1.0;
  int bf = -1, bb = 0, bdl = 0;
  long long blg = 0, blh = 0;
  // CPU semantics: dl=1 candidates evaluated first over all (f,b), dl=0
  // overrides only with strictly greater gain. Within one dl, first
  // occurrence (lowest feature) wins ties.
  for (int want_dl = 1; want_dl >= 0; --want_dl) {
    for (int f = 0; f < F; ++f) {
      int64_t kf = (int64_t)k * F + f;
      if ((int)kf_dl[kf] != want_dl) continue;
      if (kf_gain[kf] > bg) {
        bg = kf_gain[kf];
        bf = f; bb = kf_bin[kf]; bdl = want_dl;
        blg = kf_lg[kf]; blh = kf_lh[kf];
      }
    }
  }
  long long* row = out_packed + (size_t)k * 6;
  float gf = (float)bg;
  row[0] = (long long)(int)__float_as_int(gf);
  row[1] = bf;
  row[2] = bb;
  row[3] = bdl;
  row[4] = blg;
  row[5] = blh;
}

// ---------------------------------------------------------------------------
// partition_rows: stable two-pass partition of each node's ridx segment.
// Pass 1 counts left-rows per 256-row block; host does a cumsum; pass 2
// scatters with an in-block ballot prefix. Stability preserved because
// blocks are processed in segment order and lanes in row order.
// ---------------------------------------------------------------------------
#define PART_THREADS 256
#define PART_ROWS_PER_THREAD 8
#define PART_CHUNK (PART_THREADS * PART_ROWS_PER_THREAD)
#define PART_WAVES (PART_THREADS / WAVE)

// Pass 1: evaluate the split predicate ONCE per row (a ~64B cache-line
// gather from the binned matrix), cache it as one flag byte per row, and
// count per-256-row-block lefts. Pass 2 then re-reads 1 B/row instead of
// repeating the gather - halves the partition's memory cost.
// meta layout (both modes): [starts Ks | counts Ks | chunk_off Ks+1 |
// feat Ks | bin Ks | dl Ks]. Legacy: Ks = K (host-staged meta),
// limits == nullptr. Device mode: limits = {n_split, total_chunks}
// written by plan_partition_kernel; grids are BOUNDS and excess
// workgroups exit here.
__global__ void partition_count_kernel(
    const uint8_t* __restrict__ bins, const int32_t* __restrict__ ridx,
    const int64_t* __restrict__ meta,
    const int64_t* __restrict__ limits,  // nullptr or [2]
    int32_t* __restrict__ block_counts,  // [total_chunks]
    uint8_t* __restrict__ flags,         // [n] go-left per segment position
    int K, int64_t row_stride,
    const uint8_t* __restrict__ bins_T,  // [F, n] column-major copy or null
    int64_t n_rows_total) {
  int wg = blockIdx.x;
  int64_t Ks = K;
  if (limits != nullptr) {
    if (wg >= limits[1]) return;
    Ks = limits[0];
  }
  const int64_t* node_start = meta;            // starts, counts at [Ks..2Ks)
  const int64_t* chunk_off = meta + 2 * Ks;
  const int64_t* split_feat = meta + 3 * Ks + 1;
  const int64_t* split_bin = meta + 4 * Ks + 1;
  const int64_t* default_left = meta + 5 * Ks + 1;
  int lo = 0, hi = (int)Ks;
  while (lo + 1 < hi) {
    int m = (lo + hi) >> 1;
    if (chunk_off[m] <= wg) lo = m; else hi = m;
  }
  const int node = lo;
  const int64_t chunk_in_node = wg - chunk_off[node];
  const int64_t row_lo = chunk_in_node * PART_CHUNK;
  const int64_t count = node_start[Ks + node];
  const int64_t seg_start = node_start[node];
  const int feat = (int)split_feat[node], sbin = (int)split_bin[node];
  const int dl = (int)default_left[node];

  // preload all stripes' row indices + bin gathers so 8 independent
  // random-latency loads are in flight per thread (the kernel was 86%
  // memory-parked at one row per thread), accumulate the per-wave ballot
  // popcounts in a register, and reduce with ONE barrier at the end (the
  // per-stripe sync ping-pong serialized the old 4-stripe version).
  uint8_t bv[PART_ROWS_PER_THREAD];
  bool valid[PART_ROWS_PER_THREAD];
  // column-major copy: the gather touches a dense per-feature column
  // (ridx ascending within a segment -> near-sequential lines) instead
  // of one 64 B row-major line per row (64x the line traffic).
  const uint8_t* colbase =
      bins_T ? bins_T + (int64_t)feat * n_rows_total : nullptr;
  #pragma unroll
  for (int sstripe = 0; sstripe < PART_ROWS_PER_THREAD; ++sstripe) {
    const int64_t i = row_lo + sstripe * PART_THREADS + threadIdx.x;
    valid[sstripe] = i < count;
    uint64_t r = valid[sstripe] ? (uint32_t)ridx[seg_start + i] : 0;
    bv[sstripe] = colbase ? colbase[r]
                          : bins[r * (uint64_t)row_stride + feat];
  }
  __shared__ int wave_sums[PART_WAVES];
  const int wave_id = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);
  int wsum = 0;
  #pragma unroll
  for (int sstripe = 0; sstripe < PART_ROWS_PER_THREAD; ++sstripe) {
    const int64_t i = row_lo + sstripe * PART_THREADS + threadIdx.x;
    bool flag = false;
    if (valid[sstripe]) {
      const int b = bv[sstripe];
      flag = (b == 255) ? (dl != 0) : (b <= sbin);
      flags[seg_start + i] = flag ? 1 : 0;
    }
    wsum += __popcll(__ballot(flag));
  }
  if (lane == 0) wave_sums[wave_id] = wsum;
  __syncthreads();
  if (threadIdx.x == 0) {
    int total = 0;
    for (int w = 0; w < PART_WAVES; ++w) total += wave_sums[w];
    block_counts[wg] = total;
  }
}


// Per-node exclusive prefix of per-chunk left counts, on device: one WG
// per node walks its chunk range in 256-wide blocks with an LDS
// Hillis-Steele scan. Replaces a 100 KB D2H pull + host loop + 200 KB
// H2D push per depth (and their stream syncs) with one kernel and a
// K-element total pull.
__global__ void partition_prefix_kernel(
    const int32_t* __restrict__ block_counts,  // [total_chunks]
    const int64_t* __restrict__ meta,          // layout as count kernel
    const int64_t* __restrict__ limits,        // nullptr or [2]
    int64_t* __restrict__ left_before,         // [total_chunks]
    int64_t* __restrict__ node_left_total,     // [K]
    int K) {
  const int k = blockIdx.x;
  int64_t Ks = K;
  if (limits != nullptr) Ks = limits[0];
  if (k >= Ks) return;
  const int64_t* chunk_off = meta + 2 * Ks;
  const int64_t c0 = chunk_off[k], c1 = chunk_off[k + 1];
  __shared__ int64_t sc[256];
  __shared__ int64_t sc2[256];
  int64_t run = 0;
  for (int64_t base = c0; base < c1; base += 256) {
    const int nin = (int)min((int64_t)256, c1 - base);
    const int64_t v =
        threadIdx.x < nin ? (int64_t)block_counts[base + threadIdx.x] : 0;
    sc[threadIdx.x] = v;
    __syncthreads();
    // inclusive Hillis-Steele scan over 256 entries (LDS ping-pong)
    int64_t* src = sc;
    int64_t* dst = sc2;
    for (int off = 1; off < 256; off <<= 1) {
      const int64_t x = src[threadIdx.x];
      dst[threadIdx.x] =
          threadIdx.x >= off ? x + src[threadIdx.x - off] : x;
      __syncthreads();
      int64_t* t = src; src = dst; dst = t;
    }
    if (threadIdx.x < nin)
      left_before[base + threadIdx.x] = run + src[threadIdx.x] - v;
    run += src[255];
    __syncthreads();
  }
  if (threadIdx.x == 0) node_left_total[k] = run;
}

__global__ void partition_scatter_kernel(
    const uint8_t* __restrict__ flags, const int32_t* __restrict__ ridx,
    int32_t* __restrict__ ridx_out,
    const int2* __restrict__ gseg,      // segment-ordered gradient pairs
    int2* __restrict__ gseg_out,        // permuted alongside ridx
    const int64_t* __restrict__ meta,   // layout as count kernel
    const int64_t* __restrict__ limits,  // nullptr or [2]
    const int64_t* __restrict__ left_before,   // [total_chunks] excl. prefix within node
    const int64_t* __restrict__ node_left_total,  // [K]
    int K) {
  int wg = blockIdx.x;
  int64_t Ks = K;
  if (limits != nullptr) {
    if (wg >= limits[1]) return;
    Ks = limits[0];
  }
  const int64_t* node_start = meta;
  const int64_t* chunk_off = meta + 2 * Ks;
  int lo = 0, hi = (int)Ks;
  while (lo + 1 < hi) {
    int m = (lo + hi) >> 1;
    if (chunk_off[m] <= wg) lo = m; else hi = m;
  }
  const int node = lo;
  const int64_t chunk_in_node = wg - chunk_off[node];
  const int64_t row_lo = chunk_in_node * PART_CHUNK;
  const int64_t count = node_start[Ks + node];
  const int64_t seg_start = node_start[node];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave_id = threadIdx.x / WAVE;

  // Load every stripe's row data upfront (3 coalesced streams, all in
  // flight), publish every (stripe, wave) ballot popcount to LDS, ONE
  // barrier, then compute each row's exclusive left-prefix from the LDS
  // table and write. No cross-stripe serialization.
  int32_t rv[PART_ROWS_PER_THREAD];
  int2 gv[PART_ROWS_PER_THREAD];
  bool fl[PART_ROWS_PER_THREAD];
  bool valid[PART_ROWS_PER_THREAD];
  unsigned long long masks[PART_ROWS_PER_THREAD];
  __shared__ int wave_sums[PART_ROWS_PER_THREAD][PART_WAVES];
  #pragma unroll
  for (int s = 0; s < PART_ROWS_PER_THREAD; ++s) {
    const int64_t i = row_lo + s * PART_THREADS + threadIdx.x;
    valid[s] = i < count;
    rv[s] = 0;
    gv[s] = {0, 0};
    fl[s] = false;
    if (valid[s]) {
      rv[s] = ridx[seg_start + i];
      if (gseg != nullptr) gv[s] = gseg[seg_start + i];
      fl[s] = flags[seg_start + i] != 0;
    }
    masks[s] = __ballot(fl[s]);
    if (lane == 0) wave_sums[s][wave_id] = __popcll(masks[s]);
  }
  __syncthreads();

  int stripe_base = 0;  // lefts in earlier stripes of this chunk
  #pragma unroll
  for (int s = 0; s < PART_ROWS_PER_THREAD; ++s) {
    const int64_t i = row_lo + s * PART_THREADS + threadIdx.x;
    int wave_left_before = stripe_base;
    for (int w = 0; w < wave_id; ++w) wave_left_before += wave_sums[s][w];
    const int prefix_in_wave = __popcll(
        masks[s] & ((lane == 0) ? 0ull : ((~0ull) >> (64 - lane))));
    if (valid[s]) {
      // lefts strictly before row i in the whole node segment: earlier
      // chunks + earlier stripes of this chunk + this stripe's ballot
      const int64_t my_left_prefix =
          left_before[wg] + wave_left_before + prefix_in_wave;
      const int64_t pos = fl[s]
                              ? my_left_prefix
                              : node_left_total[node] + (i - my_left_prefix);
      ridx_out[seg_start + pos] = rv[s];
      if (gseg_out != nullptr) gseg_out[seg_start + pos] = gv[s];
    }
    for (int w = 0; w < PART_WAVES; ++w) stripe_base += wave_sums[s][w];
  }
}

// ---------------------------------------------------------------------------
// predict_trees: one thread per row, sequential over trees.
// ---------------------------------------------------------------------------
// Packed node: one 16-byte uint4 per node -> a tree-walk step is ONE
// vector load (vs 4 scattered loads of feat/thr/left/default_left).
// x = feature | (default_left << 30) | (is_leaf << 31)
// y = threshold bits (or leaf value bits), z = left child, w = unused.
__global__ void pack_nodes_kernel(
    const int32_t* __restrict__ feat, const float* __restrict__ thr,
    const int32_t* __restrict__ left, const uint8_t* __restrict__ default_left,
    const float* __restrict__ value, uint4* __restrict__ packed, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  int f = feat[i];
  uint4 p;
  if (f < 0) {
    p.x = 0x80000000u;
    p.y = __float_as_uint(value[i]);
    p.z = 0;
  } else {
    p.x = (uint32_t)f | ((default_left[i] != 0) ? 0x40000000u : 0u);
    p.y = __float_as_uint(thr[i]);
    p.z = (uint32_t)left[i];
  }
  p.w = 0;
  packed[i] = p;
}

// R consecutive rows per thread: their tree walks interleave, so R
// independent node loads are in flight per step (the walk is L2-latency
// bound; one chain per thread leaves the memory pipe idle).
template <int R>
__global__ void predict_trees_kernel(
    const float* __restrict__ X, const uint4* __restrict__ nodes,
    const int32_t* __restrict__ tree_ptr, float* __restrict__ out,
    float tree_weight, int64_t n, int F, int T) {
  int64_t grp = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t gstride = (int64_t)gridDim.x * blockDim.x;
  const int64_t n_grp = (n + R - 1) / R;
  for (; grp < n_grp; grp += gstride) {
    const int64_t i0 = grp * R;
    float acc[R];
    bool valid[R];
    #pragma unroll
    for (int r = 0; r < R; ++r) {
      acc[r] = 0.0f;
      valid[r] = (i0 + r) < n;
    }
    for (int t = 0; t < T; ++t) {
      const int base = tree_ptr[t];
      uint4 p[R];
      bool act[R];
      #pragma unroll
      for (int r = 0; r < R; ++r) {
        act[r] = valid[r];
        p[r] = nodes[base];
      }
      bool any = true;
      while (any) {
        any = false;
        #pragma unroll
        for (int r = 0; r < R; ++r) {
          if (!act[r]) continue;
          if (p[r].x & 0x80000000u) {
            acc[r] += __uint_as_float(p[r].y);
            act[r] = false;
          } else {
            const float v = X[(i0 + r) * (int64_t)F +
                              (p[r].x & 0x3FFFFFFFu)];
            const bool goleft =
                isnan(v) ? ((p[r].x & 0x40000000u) != 0)
                         : (v < __uint_as_float(p[r].y));
            p[r] = nodes[base + (int)p[r].z + (goleft ? 0 : 1)];
            any = true;
          }
        }
      }
    }
    #pragma unroll
    for (int r = 0; r < R; ++r) {
      if (valid[r]) out[i0 + r] += acc[r] * tree_weight;
    }
  }
}

// ---------------------------------------------------------------------------
// update_margins: margin[ridx[seg]] += leaf_val[node]
// ---------------------------------------------------------------------------
// 8 rows per thread: the margin update is a random-line RMW per row
// (leaf rows are ~n_leaves apart in the dense margin vector); one row
// per thread left the kernel latency-parked at 3.4x the write floor.
#define MARGIN_ROWS_PER_THREAD 8
#define MARGIN_CHUNK (PART_THREADS * MARGIN_ROWS_PER_THREAD)
__global__ void update_margins_kernel(
    float* __restrict__ margin, const int32_t* __restrict__ ridx,
    const int32_t* __restrict__ ridx_b,      // second ping-pong buffer
    const int64_t* __restrict__ node_start,  // [K] + counts at [K..2K)
    const int64_t* __restrict__ chunk_off,   // [K+1]
    const int64_t* __restrict__ parity,      // [K] 0 -> ridx, 1 -> ridx_b
    const float* __restrict__ leaf_vals, int K) {
  int wg = blockIdx.x;
  int lo = 0, hi = K;
  while (lo + 1 < hi) {
    int m = (lo + hi) >> 1;
    if (chunk_off[m] <= wg) lo = m; else hi = m;
  }
  const int node = lo;
  const int64_t chunk_in_node = wg - chunk_off[node];
  const int64_t row_lo = chunk_in_node * MARGIN_CHUNK;
  const int64_t count = node_start[K + node];
  const int64_t seg_start = node_start[node];
  if (parity != nullptr && parity[node]) ridx = ridx_b;
  const float v = leaf_vals[node];
  int32_t rv[MARGIN_ROWS_PER_THREAD];
  bool valid[MARGIN_ROWS_PER_THREAD];
  #pragma unroll
  for (int s = 0; s < MARGIN_ROWS_PER_THREAD; ++s) {
    const int64_t i = row_lo + s * PART_THREADS + threadIdx.x;
    valid[s] = i < count;
    rv[s] = valid[s] ? ridx[seg_start + i] : 0;
  }
  #pragma unroll
  for (int s = 0; s < MARGIN_ROWS_PER_THREAD; ++s) {
    if (valid[s]) margin[(uint32_t)rv[s]] += v;
  }
}

// ---------------------------------------------------------------------------
// lambdarank_grad: pairwise LambdaRank gradients, one workgroup per query
// group; each thread owns docs i = tid, tid+256, ... and loops all other
// docs j sequentially, so per-doc accumulation order is fixed ->
// deterministic across runs (no atomics). NDCG variant weights each pair
// by |delta NDCG| from precomputed ranks + idcg.
// ---------------------------------------------------------------------------
__global__ void lambdarank_kernel(
    const float* __restrict__ margin, const float* __restrict__ label,
    const int64_t* __restrict__ group_ptr,  // [G+1]
    const int32_t* __restrict__ rank,       // [n] rank within group
    const double* __restrict__ idcg,        // [G] (ndcg mode) or nullptr
    float2* __restrict__ out,               // [n] (grad, hess)
    int use_ndcg) {
  const int g = blockIdx.x;
  const int64_t s0 = group_ptr[g], s1 = group_ptr[g + 1];
  const int len = (int)(s1 - s0);
  const double inv_idcg =
      (use_ndcg && idcg[g] > 0.0) ? 1.0 / idcg[g] : 0.0;
  if (use_ndcg && inv_idcg == 0.0) {
    for (int i = threadIdx.x; i < len; i += blockDim.x)
      out[s0 + i] = make_float2(0.f, 0.f);
    return;
  }
  for (int i = threadIdx.x; i < len; i += blockDim.x) {
    const float yi = label[s0 + i];
    const float mi = margin[s0 + i];
    const double gain_i = use_ndcg ? (exp2((double)yi) - 1.0) : 0.0;
    const double disc_i =
        use_ndcg ? 1.0 / log2((double)rank[s0 + i] + 2.0) : 0.0;
    double gacc = 0.0, hacc = 0.0;
    for (int j = 0; j < len; ++j) {
      const float yj = label[s0 + j];
      if (yj == yi) continue;
      const float mj = margin[s0 + j];
      double w = 1.0;
      if (use_ndcg) {
        const double gain_j = exp2((double)yj) - 1.0;
        const double disc_j = 1.0 / log2((double)rank[s0 + j] + 2.0);
        w = fabs((gain_i - gain_j) * (disc_i - disc_j)) * inv_idcg;
      }
      if (yi > yj) {
        // i should rank above j: pair (i, j)
        const double rho = 1.0 / (1.0 + exp((double)(mi - mj)));
        double hij = rho * (1.0 - rho);
        if (hij < 1e-16) hij = 1e-16;
        gacc += -rho * w;
        hacc += hij * w;
      } else {
        // j should rank above i: pair (j, i), i is the loser
        const double rho = 1.0 / (1.0 + exp((double)(mj - mi)));
        double hij = rho * (1.0 - rho);
        if (hij < 1e-16) hij = 1e-16;
        gacc += rho * w;
        hacc += hij * w;
      }
    }
    out[s0 + i] = make_float2((float)gacc, (float)hacc);
  }
}

// ===========================================================================
// Host-side launchers / bindings
// ===========================================================================

static torch::Tensor make_chunk_off_cpu(const torch::Tensor& counts_cpu,
                                        int64_t rows_per_chunk,
                                        int64_t* total) {
  const int64_t K = counts_cpu.size(0);
  auto off = torch::zeros({K + 1}, torch::kInt64);
  auto acc = off.accessor<int64_t, 1>();
  auto cacc = counts_cpu.accessor<int64_t, 1>();
  for (int64_t k = 0; k < K; ++k) {
    acc[k + 1] = acc[k] + (cacc[k] + rows_per_chunk - 1) / rows_per_chunk;
  }
  *total = acc[K];
  return off;
}

static torch::Tensor cat_start_count(const torch::Tensor& starts_cpu,
                                     const torch::Tensor& counts_cpu,
                                     torch::Device dev) {
  // layout used by kernels: [0..K) starts, [K..2K) counts
  return torch::cat({starts_cpu, counts_cpu}).to(dev);
}

torch::Tensor quantize_gpair(torch::Tensor gpair, double scale_g, double scale_h) {
  TORCH_CHECK(on_gpu(gpair) && gpair.dtype() == torch::kFloat32);
  auto out = torch::empty_like(gpair, gpair.options().dtype(torch::kInt32));
  int64_t n = gpair.size(0);
  if (n == 0) return out;
  auto stream = c10::hip::getCurrentHIPStream();
  int64_t blocks = std::min<int64_t>(ceil_div(n, 256), 8192);
  hipLaunchKernelGGL(quantize_gpair_kernel, dim3(blocks), dim3(256), 0,
                     stream.stream(), (const float2*)gpair.data_ptr<float>(),
                     (int2*)out.data_ptr<int32_t>(), scale_g, scale_h, n);
  return out;
}

torch::Tensor bin_matrix(torch::Tensor values, torch::Tensor cuts_flat,
                         torch::Tensor cut_ptr) {
  TORCH_CHECK(on_gpu(values) && values.dim() == 2);
  int64_t n = values.size(0);
  int F = (int)values.size(1);
  // padded allocation: 16B-aligned row stride, pad bytes = 255 (missing)
  // so the vectorized histogram path can read whole uint4 feature blocks
  int64_t F_pad = ((int64_t)F + 15) / 16 * 16;
  auto full = torch::full({n, F_pad}, 255,
                          values.options().dtype(torch::kUInt8));
  auto out = full.narrow(1, 0, F);
  if (n == 0) return out;
  int total_cuts = (int)cuts_flat.size(0);
  auto stream = c10::hip::getCurrentHIPStream();
  int64_t blocks = std::min<int64_t>(ceil_div(n * F, 256), 16384);
  size_t lds = (size_t)total_cuts * sizeof(float);
  if (lds <= 64 * 1024) {
    hipLaunchKernelGGL((bin_matrix_kernel<true>), dim3(blocks), dim3(256), lds,
                       stream.stream(), values.data_ptr<float>(),
                       full.data_ptr<uint8_t>(), cuts_flat.data_ptr<float>(),
                       cut_ptr.data_ptr<int64_t>(), n, F, total_cuts, F_pad);
  } else {
    hipLaunchKernelGGL((bin_matrix_kernel<false>), dim3(blocks), dim3(256), 0,
                       stream.stream(), values.data_ptr<float>(),
                       full.data_ptr<uint8_t>(), cuts_flat.data_ptr<float>(),
                       cut_ptr.data_ptr<int64_t>(), n, F, total_cuts, F_pad);
  }
  return out;
}


// ---------------------------------------------------------------------------
// Pinned staging: pageable H2D/D2H around every per-depth launch blocks the
// host in the driver's staging path. Each call site owns a cached pinned
// buffer; an event guards against overwriting a buffer whose async H2D is
// still in flight (can happen in the chunked-overlap histogram path where
// several builds are enqueued back-to-back with no intervening sync).
// ---------------------------------------------------------------------------
struct PinnedStager {
  torch::Tensor buf;
  hipEvent_t ev = nullptr;
  bool pending = false;
  torch::Tensor get(int64_t n, torch::Dtype dt = torch::kInt64) {
    if (pending) {
      (void)hipEventSynchronize(ev);
      pending = false;
    }
    if (!buf.defined() || buf.numel() < n || buf.scalar_type() != dt) {
      int64_t cap = 4096;
      while (cap < n) cap *= 2;
      buf = torch::empty({cap},
                         torch::TensorOptions().dtype(dt).pinned_memory(true));
    }
    return buf.narrow(0, 0, n);
  }
  void mark(hipStream_t s) {
    if (!ev) (void)hipEventCreateWithFlags(&ev, hipEventDisableTiming);
    (void)hipEventRecord(ev, s);
    pending = true;
  }
};

// cat into a pinned stager view. The view's length MUST equal the summed
// input length: torch::cat_out resizes a mismatched output, and the
// resized allocation is pageable — silently defeating the pinned staging
// (and flooding stderr with Resize.cpp warnings). Checked, not assumed.
static void cat_into_pinned(torch::Tensor out,
                            std::vector<torch::Tensor> parts) {
  int64_t total = 0;
  for (auto& p : parts) total += p.numel();
  TORCH_CHECK(out.numel() == total, "pinned stager size mismatch: have ",
              out.numel(), " need ", total);
  TORCH_CHECK(out.is_pinned(), "stager buffer lost pinning");
  torch::cat_out(out, parts);
}

torch::Tensor build_histogram(torch::Tensor bins, torch::Tensor gpair_q,
                              torch::Tensor ridx, torch::Tensor starts,
                              torch::Tensor counts, int64_t n_bins,
                              int64_t f_lo, int64_t f_hi,
                              torch::Tensor hist, bool pregathered) {
  // builds features [f_lo, f_hi) into the caller-provided [K, F, n_bins, 2]
  // histogram (zeroed by the caller); ranges let the driver overlap the
  // RCCL AllReduce of one feature block with the build of the next.
  TORCH_CHECK(on_gpu(bins) && bins.dtype() == torch::kUInt8);
  const int K = (int)starts.size(0);
  const int F = (int)bins.size(1);
  auto dev = bins.device();
  if (K == 0) return hist;
  auto starts_cpu = starts.to(torch::kCPU).to(torch::kInt64);
  auto counts_cpu = counts.to(torch::kCPU).to(torch::kInt64);

  // feature-block config decides the kernel flavor, which decides the
  // row-chunk size, so resolve it before computing chunk offsets.
  const int64_t row_stride0 = bins.stride(0);
  const bool vec16_pre = (row_stride0 % 16 == 0) && n_bins <= 256;
  int fb_pre = 16;
  if (const char* e = getenv("RXGB_HIST_FB")) {
    if (atoi(e) == 8) fb_pre = 8;
  }
  const int n_fb_pre =
      vec16_pre ? (int)ceil_div(f_hi - f_lo, (int64_t)fb_pre) : 1;
  // multifb: one WG sweeps all feature blocks with register-cached
  // gpairs/ridx (see kernel comment). Default on for multi-block ranges;
  // RXGB_HIST_MULTIFB=0 disables, RXGB_HIST_MULTIFB_R in {8,16,32}.
  // any multi-block range: with the SoA LDS layout + interleaved merge
  // it wins at 2 blocks too (HIGGS 8.18 vs 8.31 ms/round) and by 11% at
  // 13 blocks (100M x 200: 138.7 vs 155.6 ms/round at R=8).
  bool multifb = vec16_pre && fb_pre == 16 && n_fb_pre >= 2;
  if (const char* e = getenv("RXGB_HIST_MULTIFB")) {
    if (atoi(e) == 0) multifb = false;
  }
  // Default mode for multi-block ranges: XCD-swizzled co-resident grid
  // (hist 102 -> 56 ms at 100M x 200; HIGGS 8.31 -> 7.25 ms/round).
  // RXGB_HIST_MODE=multifb selects the register-cached block-sweep
  // variant instead; =base the original (chunk, block) 2-D grid.
  bool xcd_mode = multifb;
  if (const char* e = getenv("RXGB_HIST_MODE")) {
    if (strcmp(e, "multifb") == 0) xcd_mode = false;
    else if (strcmp(e, "base") == 0) { xcd_mode = false; multifb = false; }
  }
  if (xcd_mode) multifb = false;
  // depth-aware chunk size: large frontiers (deep depths) have small
  // per-node segments where 8192-row chunks load-balance better
  // (measured d12 6.81 vs 6.84 ms/round); shallow depths keep 16384
  // (fewer LDS tile merges per node; 16384 won the 100M sweep)
  int xcd_rows = K >= 512 ? 8192 : 16384;
  if (const char* e = getenv("RXGB_HIST_XCD_ROWS")) {
    int v = atoi(e);
    if (v >= 512 && v <= 65536) xcd_rows = v;
  }
  // R=8 (4096 rows/WG) measured best at 100M x 200: finer chunks load-
  // balance deep depths better than R=16/32, and occupancy is LDS-bound.
  // tiny-node threshold for the direct-to-global path. Default OFF:
  // measured on HIGGS-11Mx28 depth 12, direct=512/1024/2048 gave
  // 11.84/12.14/12.88 ms/round vs 11.65 with the LDS route - the
  // skip-if-zero merge already makes empty tiles cheap, and the direct
  // path's global-atomic latency is not hidden at 256-thread occupancy.
  // Kept as an env knob for wider matrices / other shapes.
  int direct_rows = 0;
  if (const char* e = getenv("RXGB_HIST_DIRECT_ROWS")) {
    int v = atoi(e);
    if (v >= 0 && v <= 65536) direct_rows = v;
  }
  int mfb_r = 8;
  if (const char* e = getenv("RXGB_HIST_MULTIFB_R")) {
    int v = atoi(e);
    if (v == 4 || v == 8 || v == 16 || v == 32) mfb_r = v;
  }
  const int rows_per_wg = multifb ? mfb_r * HIST_THREADS
                          : xcd_mode ? xcd_rows
                                     : HIST_ROWS_PER_WG;

  int64_t total_chunks = 0;
  auto chunk_off_cpu =
      make_chunk_off_cpu(counts_cpu, rows_per_wg, &total_chunks);
  if (total_chunks == 0) return hist;

  // gather gpairs into segment order once (coalesced hist reads)
  auto starts_acc = starts_cpu.accessor<int64_t, 1>();
  auto counts_acc = counts_cpu.accessor<int64_t, 1>();
  // Segments are contiguous [start, start+count) spans of ridx; gather the
  // covering range [min_start, max_end) in one kernel.
  int64_t min_start = INT64_MAX, max_end = 0;
  for (int k = 0; k < K; ++k) {
    min_start = std::min(min_start, starts_acc[k]);
    max_end = std::max(max_end, starts_acc[k] + counts_acc[k]);
  }
  if (min_start == INT64_MAX) { min_start = 0; max_end = 0; }
  auto stream = c10::hip::getCurrentHIPStream();
  int64_t span = max_end - min_start;
  TORCH_CHECK(gpair_q.dtype() == torch::kInt32,
              "gpair_q must be int32 packed pairs");
  torch::Tensor gpair_seg;
  if (pregathered) {
    // gpair_q is ALREADY in segment order aligned with ridx (the
    // partition scatter permutes it) - no gather pass needed
    gpair_seg = gpair_q;
    min_start = 0;
  } else {
    gpair_seg = torch::empty({std::max<int64_t>(span, 1), 2},
                             gpair_q.options());
    if (span > 0) {
      int64_t blocks = std::min<int64_t>(ceil_div(span, 256), 8192);
      hipLaunchKernelGGL(gather_gpair_kernel, dim3(blocks), dim3(256), 0,
                         stream.stream(),
                         (const int2*)gpair_q.data_ptr<int32_t>(),
                         ridx.data_ptr<int32_t>() + min_start,
                         (int2*)gpair_seg.data_ptr<int32_t>(), span);
    }
  }
  // ONE H2D copy for all control data: [starts_adj(K) | counts(K) |
  // chunk_off(K+1)] - tiny pageable copies around kernel launches were
  // the training loop's dominant host cost.
  auto starts_adj = starts_cpu - min_start;
  static thread_local PinnedStager hist_meta_stager;
  auto meta_cpu = hist_meta_stager.get(3 * K + 1);
  cat_into_pinned(meta_cpu, {starts_adj, counts_cpu, chunk_off_cpu});
  auto meta = meta_cpu.to(dev, /*non_blocking=*/true);
  hist_meta_stager.mark(stream.stream());
  int64_t* mp = meta.data_ptr<int64_t>();
  int64_t* sc_adj_p = mp;          // starts at [0..K), counts at [K..2K)
  int64_t* chunk_off_p = mp + 2 * K;

  // feature-block size: fit the LDS tile (fb * n_bins * 16 B) within the
  // 64 KiB dynamic-LDS default so two workgroups co-reside per CU.
  // Padded (16B-aligned row stride) matrices take the vectorized path:
  // fb = 16 features per uint4 row load.
  const int64_t row_stride = row_stride0;
  const bool vec16 = vec16_pre;
  int fb_size = vec16
                    ? 16
                    : (int)std::min<int64_t>(F, (64 * 1024) / (n_bins * 16));
  if (fb_size < 1) fb_size = 1;
  // experiment knob: RXGB_HIST_FB=8 halves the LDS tile (4 workgroups/CU
  // instead of 2) at the cost of 2x gradient re-reads per depth
  if (vec16 && fb_pre == 8) fb_size = 8;
  TORCH_CHECK(f_lo % fb_size == 0 && f_lo < f_hi && f_hi <= F,
              "feature range must align to the block size ", fb_size);
  const int n_fb = (int)ceil_div(f_hi - f_lo, fb_size);
  const size_t lds = (size_t)fb_size * n_bins * 2 * sizeof(long long);

  // ridx pointer offset so seg indices align with gpair_seg
  if (xcd_mode) {
    const int64_t grid = ceil_div(total_chunks, 8) * 8 * (int64_t)n_fb;
    // 1024-thread WGs double waves/SIMD (2 WGs/CU within 160 KB LDS
    // either way) and win on huge matrices (100M: hist 55.6 -> 53.9 ms)
    // but lose ~3% on 11M-row shapes (deep-depth tail waste), so gate on
    // matrix size; RXGB_HIST_XCD_THREADS=512/1024 overrides.
    int xt = bins.size(0) > (int64_t)32 * 1024 * 1024 ? 1024 : 512;
    if (const char* e = getenv("RXGB_HIST_XCD_THREADS")) {
      int v = atoi(e);
      if (v == 512 || v == 1024) xt = v;
    }
    auto launch_xcd = [&](auto tc) {
      hipLaunchKernelGGL((build_histogram_xcd_kernel<decltype(tc)::value>),
                         dim3((uint32_t)grid), dim3(decltype(tc)::value),
                         lds, stream.stream(), bins.data_ptr<uint8_t>(),
                         (const int2*)gpair_seg.data_ptr<int32_t>(),
                         ridx.data_ptr<int32_t>() + min_start, sc_adj_p,
                         chunk_off_p,
                         reinterpret_cast<long long*>(hist.data_ptr<int64_t>()),
                         K, F, (int)n_bins, n_fb, row_stride, (int)f_lo,
                         rows_per_wg, (int)total_chunks, direct_rows);
    };
    if (xt == 512)
      launch_xcd(std::integral_constant<int, 512>{});
    else
      launch_xcd(std::integral_constant<int, 1024>{});
  } else if (multifb) {
    auto launch_mfb = [&](auto rc) {
      hipLaunchKernelGGL((build_histogram_multifb_kernel<decltype(rc)::value>),
                         dim3((uint32_t)total_chunks), dim3(HIST_THREADS),
                         lds, stream.stream(), bins.data_ptr<uint8_t>(),
                         (const int2*)gpair_seg.data_ptr<int32_t>(),
                         ridx.data_ptr<int32_t>() + min_start, sc_adj_p,
                         chunk_off_p,
                         reinterpret_cast<long long*>(hist.data_ptr<int64_t>()),
                         K, F, (int)n_bins, n_fb, row_stride, (int)f_lo);
    };
    if (mfb_r == 4)
      launch_mfb(std::integral_constant<int, 4>{});
    else if (mfb_r == 16)
      launch_mfb(std::integral_constant<int, 16>{});
    else if (mfb_r == 32)
      launch_mfb(std::integral_constant<int, 32>{});
    else
      launch_mfb(std::integral_constant<int, 8>{});
  } else if (vec16 && fb_size == 8) {
    hipLaunchKernelGGL((build_histogram_kernel<8>),
                       dim3((uint32_t)total_chunks, n_fb),
                       dim3(HIST_THREADS), lds, stream.stream(),
                       bins.data_ptr<uint8_t>(),
                       (const int2*)gpair_seg.data_ptr<int32_t>(),
                       ridx.data_ptr<int32_t>() + min_start,
                       sc_adj_p, chunk_off_p,
                       reinterpret_cast<long long*>(hist.data_ptr<int64_t>()),
                       K, F, (int)n_bins, fb_size, row_stride, (int)f_lo,
                       rows_per_wg, direct_rows);
  } else if (vec16) {
    hipLaunchKernelGGL((build_histogram_kernel<16>),
                       dim3((uint32_t)total_chunks, n_fb),
                       dim3(HIST_THREADS), lds, stream.stream(),
                       bins.data_ptr<uint8_t>(),
                       (const int2*)gpair_seg.data_ptr<int32_t>(),
                       ridx.data_ptr<int32_t>() + min_start,
                       sc_adj_p, chunk_off_p,
                       reinterpret_cast<long long*>(hist.data_ptr<int64_t>()),
                       K, F, (int)n_bins, fb_size, row_stride, (int)f_lo,
                       rows_per_wg, direct_rows);
  } else {
    hipLaunchKernelGGL((build_histogram_kernel<0>),
                       dim3((uint32_t)total_chunks, n_fb),
                       dim3(HIST_THREADS), lds, stream.stream(),
                       bins.data_ptr<uint8_t>(),
                       (const int2*)gpair_seg.data_ptr<int32_t>(),
                       ridx.data_ptr<int32_t>() + min_start,
                       sc_adj_p, chunk_off_p,
                       reinterpret_cast<long long*>(hist.data_ptr<int64_t>()),
                       K, F, (int)n_bins, fb_size, row_stride, (int)f_lo,
                       rows_per_wg, direct_rows);
  }
  return hist;
}

std::vector<torch::Tensor> find_splits(torch::Tensor hist, torch::Tensor parent_g,
                                       torch::Tensor parent_h,
                                       torch::Tensor feat_bins, double scale_g,
                                       double scale_h, double lam, double alpha,
                                       double gamma, double mcw,
                                       torch::Tensor mono, torch::Tensor bounds,
                                       torch::Tensor allowed, bool pull) {
  const int K = (int)hist.size(0);
  const int F = (int)hist.size(1);
  const int B = (int)hist.size(2);
  auto dev = hist.device();
  auto optsd = torch::TensorOptions().dtype(torch::kFloat64).device(dev);
  auto optsi = torch::TensorOptions().dtype(torch::kInt32).device(dev);
  auto optsl = torch::TensorOptions().dtype(torch::kInt64).device(dev);
  auto optsb = torch::TensorOptions().dtype(torch::kUInt8).device(dev);
  auto kf_gain = torch::empty({(int64_t)K * F}, optsd);
  auto kf_bin = torch::empty({(int64_t)K * F}, optsi);
  auto kf_dl = torch::empty({(int64_t)K * F}, optsb);
  auto kf_lg = torch::empty({(int64_t)K * F}, optsl);
  auto kf_lh = torch::empty({(int64_t)K * F}, optsl);
  auto stream = c10::hip::getCurrentHIPStream();
  int64_t total = (int64_t)K * F;
  // one 64-lane wave per (k, f), four waves per 256-thread block; no LDS
  const int waves_per_block = 4;
  hipLaunchKernelGGL(find_splits_kf_kernel,
                     dim3((uint32_t)ceil_div(total, waves_per_block)),
                     dim3(waves_per_block * WAVE), 0,
                     stream.stream(),
                     reinterpret_cast<const long long*>(hist.data_ptr<int64_t>()),
                     reinterpret_cast<const long long*>(parent_g.data_ptr<int64_t>()),
                     reinterpret_cast<const long long*>(parent_h.data_ptr<int64_t>()),
                     feat_bins.data_ptr<int32_t>(), scale_g, scale_h, lam,
                     alpha, gamma, mcw,
                     mono.numel() ? mono.data_ptr<int8_t>() : nullptr,
                     bounds.numel() ? bounds.data_ptr<double>() : nullptr,
                     allowed.numel() ? allowed.data_ptr<uint8_t>() : nullptr,
                     kf_gain.data_ptr<double>(),
                     kf_bin.data_ptr<int32_t>(), kf_dl.data_ptr<uint8_t>(),
                     reinterpret_cast<long long*>(kf_lg.data_ptr<int64_t>()),
                     reinterpret_cast<long long*>(kf_lh.data_ptr<int64_t>()), K, F, B);
  auto out_packed = torch::empty({K, 6}, optsl);
  hipLaunchKernelGGL(find_splits_reduce_kernel, dim3((uint32_t)ceil_div(K, 64)),
                     dim3(64), 0, stream.stream(), kf_gain.data_ptr<double>(),
                     kf_bin.data_ptr<int32_t>(), kf_dl.data_ptr<uint8_t>(),
                     reinterpret_cast<const long long*>(kf_lg.data_ptr<int64_t>()),
                     reinterpret_cast<const long long*>(kf_lh.data_ptr<int64_t>()),
                     reinterpret_cast<long long*>(out_packed.data_ptr<int64_t>()),
                     K, F);
  if (!pull) return {out_packed};  // device [K,6]: consumed by the
                                   // fused partition (single-sync path)
  // pinned D2H: the caller consumes (copies out of) the result
  // immediately, so returning a view of the cached buffer is safe
  static thread_local PinnedStager split_out_stager;
  auto out_cpu = split_out_stager.get((int64_t)K * 6).view({K, 6});
  out_cpu.copy_(out_packed);
  return {out_cpu};
}


// ---------------------------------------------------------------------------
// plan_partition: ONE device kernel turns the depth's packed split
// results ([K,6]: f32-gain bits, feat, bin, dl, lg, lh) into the
// compacted partition metadata the partition kernels consume
// (starts|counts|chunk_off|feat|bin|dl of the SPLIT nodes only), so the
// partition can launch without the host ever seeing the splits. The
// split predicate (gain > 0, feat >= 0, finite) is bit-identical to the
// host replay's numpy predicate - both read the same f32 bits.
// Single workgroup: K <= 8192 and the work per node is trivial.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(1024) void plan_partition_kernel(
    const long long* __restrict__ packed,      // [K, 6]
    const int64_t* __restrict__ starts_ord,    // [K] scan-slot order
    const int64_t* __restrict__ counts_ord,    // [K]
    int64_t* __restrict__ meta,     // [6K+1] compacted partition meta
    int64_t* __restrict__ scalars,  // [2]: n_split, total_chunks
    int K) {
  __shared__ int64_t s_ok[1024];
  __shared__ int64_t s_ch[1024];
  const int per = (K + blockDim.x - 1) / blockDim.x;
  const int lo = threadIdx.x * per;
  const int hi = min(K, lo + per);
  // thread-local pass: ok flags + chunk counts
  int64_t my_ok = 0, my_ch = 0;
  for (int k = lo; k < hi; ++k) {
    const float gain = __int_as_float((int)packed[(size_t)k * 6]);
    const long long f = packed[(size_t)k * 6 + 1];
    const bool ok = (gain > 0.0f) && (f >= 0) && isfinite(gain);
    if (ok) {
      ++my_ok;
      my_ch += (counts_ord[k] + PART_CHUNK - 1) / PART_CHUNK;
    }
  }
  s_ok[threadIdx.x] = my_ok;
  s_ch[threadIdx.x] = my_ch;
  __syncthreads();
  // inclusive block scan (Hillis-Steele over 1024)
  for (int off = 1; off < 1024; off <<= 1) {
    int64_t a = 0, b = 0;
    if ((int)threadIdx.x >= off) {
      a = s_ok[threadIdx.x - off];
      b = s_ch[threadIdx.x - off];
    }
    __syncthreads();
    s_ok[threadIdx.x] += a;
    s_ch[threadIdx.x] += b;
    __syncthreads();
  }
  const int64_t n_split = s_ok[1023];
  const int64_t total_chunks = s_ch[1023];
  int64_t pos = s_ok[threadIdx.x] - my_ok;   // exclusive prefixes
  int64_t coff = s_ch[threadIdx.x] - my_ch;
  // meta layout (Ks = n_split): [starts Ks | counts Ks | chunk_off Ks+1
  //  | feat Ks | bin Ks | dl Ks] packed back-to-back at n_split-based
  // offsets; chunk_off tail padded with total_chunks so the kernels'
  // binary search over a BOUND-sized node range never maps a chunk to a
  // phantom node.
  int64_t* m_starts = meta;
  int64_t* m_counts = meta + n_split;
  int64_t* m_coff = meta + 2 * n_split;
  int64_t* m_feat = meta + 3 * n_split + 1;
  int64_t* m_bin = meta + 4 * n_split + 1;
  int64_t* m_dl = meta + 5 * n_split + 1;
  for (int k = lo; k < hi; ++k) {
    const float gain = __int_as_float((int)packed[(size_t)k * 6]);
    const long long f = packed[(size_t)k * 6 + 1];
    const bool ok = (gain > 0.0f) && (f >= 0) && isfinite(gain);
    if (ok) {
      m_starts[pos] = starts_ord[k];
      m_counts[pos] = counts_ord[k];
      m_coff[pos] = coff;
      m_feat[pos] = f;
      m_bin[pos] = packed[(size_t)k * 6 + 2];
      m_dl[pos] = packed[(size_t)k * 6 + 3];
      coff += (counts_ord[k] + PART_CHUNK - 1) / PART_CHUNK;
      ++pos;
    }
  }
  if (threadIdx.x == 1023) m_coff[n_split] = total_chunks;
  if (threadIdx.x == 0) {
    scalars[0] = n_split;
    scalars[1] = total_chunks;
  }
}

// Fused scan-consume partition: find_splits' packed output feeds the
// partition directly on device; the host gets {packed | left_counts}
// in ONE pinned D2H it reads after a single stream sync per depth
// (replacing the 2 syncs/depth of the host-planned path).
std::vector<torch::Tensor> partition_rows_from_packed(
    torch::Tensor bins, torch::Tensor ridx, torch::Tensor starts_ord,
    torch::Tensor counts_ord, torch::Tensor packed, torch::Tensor gseg,
    torch::Tensor bins_t, int64_t chunk_bound, torch::Tensor ridx_dest) {
  const int K = (int)packed.size(0);     // frontier bound (scan-slot count)
  auto dev = bins.device();
  // ping-pong destination: the caller tracks which buffer each leaf
  // segment's rows last landed in (parity) for the margin update, so
  // no full clone is needed (the 44 MB stream-ordered copy per depth
  // serialized the loop like the gseg clone did)
  auto ridx_out = ridx_dest.numel() ? ridx_dest : ridx.clone();
  // gseg needs no clone: the scatter fully rewrites every SPLIT
  // segment, and a leaf segment's gradient pairs are never read again
  // (ridx IS read for the end-of-round leaf margin update, so it keeps
  // the clone). Saves an 88 MB device copy per depth at 11M rows.
  auto gseg_out = torch::empty_like(gseg);
  auto optsl = torch::TensorOptions().dtype(torch::kInt64).device(dev);
  auto meta = torch::empty({6 * (int64_t)K + 2}, optsl);
  auto scalars = torch::zeros({2}, optsl);
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(plan_partition_kernel, dim3(1), dim3(1024), 0,
                     stream.stream(),
                     reinterpret_cast<const long long*>(packed.data_ptr<int64_t>()),
                     starts_ord.data_ptr<int64_t>(),
                     counts_ord.data_ptr<int64_t>(),
                     meta.data_ptr<int64_t>(), scalars.data_ptr<int64_t>(),
                     K);
  int64_t* mp = meta.data_ptr<int64_t>();
  const int64_t* sc2 = scalars.data_ptr<int64_t>();
  auto block_counts = torch::empty({std::max<int64_t>(chunk_bound, 1)},
      torch::TensorOptions().dtype(torch::kInt32).device(dev));
  auto flags = torch::empty({ridx.size(0)},
      torch::TensorOptions().dtype(torch::kUInt8).device(dev));
  // device-limit form: node/chunk layout pointers are n_split-based
  // inside `meta`, discovered by each kernel from scalars[0]
  hipLaunchKernelGGL(partition_count_kernel,
                     dim3((uint32_t)std::max<int64_t>(chunk_bound, 1)),
                     dim3(PART_THREADS), 0, stream.stream(),
                     bins.data_ptr<uint8_t>(), ridx.data_ptr<int32_t>(),
                     mp, sc2, block_counts.data_ptr<int32_t>(),
                     flags.data_ptr<uint8_t>(), K, bins.stride(0),
                     bins_t.numel() ? bins_t.data_ptr<uint8_t>() : nullptr,
                     bins.size(0));
  auto left_before = torch::empty({std::max<int64_t>(chunk_bound, 1)}, optsl);
  auto node_left_total = torch::zeros({K}, optsl);
  hipLaunchKernelGGL(partition_prefix_kernel,
                     dim3((uint32_t)std::max(K, 1)), dim3(256), 0,
                     stream.stream(),
                     block_counts.data_ptr<int32_t>(), mp, sc2,
                     left_before.data_ptr<int64_t>(),
                     node_left_total.data_ptr<int64_t>(), K);
  // ONE async pinned D2H of everything the host replay needs -
  // enqueued BEFORE the scatter, so the host's single sync wakes as
  // soon as scan+plan+count+prefix are done and ALL its bookkeeping
  // overlaps the scatter (the first version synced past the scatter,
  // which serialized the bookkeeping after it and measured slower
  // than the 2-sync structure)
  static thread_local PinnedStager pull_stager;
  auto pull = pull_stager.get(7 * (int64_t)K);
  pull.narrow(0, 0, 6 * (int64_t)K)
      .copy_(packed.view({-1}), /*non_blocking=*/true);
  pull.narrow(0, 6 * (int64_t)K, K)
      .copy_(node_left_total, /*non_blocking=*/true);
  pull_stager.mark(stream.stream());
  auto cb = torch::full({1}, chunk_bound, torch::kInt64);
  return {ridx_out, gseg_out, pull, meta, scalars, left_before,
          node_left_total, flags, cb};
}

// Scatter phase of the device-planned partition: launched AFTER the
// caller records its pull event, so the host's event-wait wakes before
// the scatter and all bookkeeping overlaps it.
void partition_scatter_from_packed(
    torch::Tensor ridx, torch::Tensor ridx_out, torch::Tensor gseg,
    torch::Tensor gseg_out, torch::Tensor meta, torch::Tensor scalars,
    torch::Tensor left_before, torch::Tensor node_left_total,
    torch::Tensor flags, torch::Tensor cb) {
  const int K = (int)node_left_total.size(0);
  const int64_t chunk_bound = cb.item<int64_t>();
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(partition_scatter_kernel,
                     dim3((uint32_t)std::max<int64_t>(chunk_bound, 1)),
                     dim3(PART_THREADS), 0, stream.stream(),
                     flags.data_ptr<uint8_t>(), ridx.data_ptr<int32_t>(),
                     ridx_out.data_ptr<int32_t>(),
                     gseg.numel() ? (const int2*)gseg.data_ptr<int32_t>()
                                  : nullptr,
                     gseg_out.numel() ? (int2*)gseg_out.data_ptr<int32_t>()
                                      : nullptr,
                     meta.data_ptr<int64_t>(), scalars.data_ptr<int64_t>(),
                     left_before.data_ptr<int64_t>(),
                     node_left_total.data_ptr<int64_t>(), K);
}

// Two-phase partition: `begin` launches count+prefix and returns
// immediately so the caller's host-side tree bookkeeping overlaps those
// kernels; `finish` pulls the per-node left totals (the one host sync)
// and launches the scatter (which the caller's subsequent bookkeeping
// then overlaps). partition_rows keeps the one-call form.
std::vector<torch::Tensor> partition_rows_begin(
    torch::Tensor bins, torch::Tensor ridx, torch::Tensor starts,
    torch::Tensor counts, torch::Tensor split_feat,
    torch::Tensor split_bin, torch::Tensor default_left,
    torch::Tensor gseg, torch::Tensor bins_t, bool gseg_full_rewrite) {
  const int K = (int)starts.size(0);
  auto dev = bins.device();
  auto ridx_out = ridx.clone();
  // Depthwise growth rewrites every still-live gseg segment each depth,
  // so its output needs no clone (see partition_rows_from_packed).
  // Lossguide partitions ONE node per expansion and keeps reading the
  // other candidates''' segments -> it must clone.
  auto gseg_out =
      gseg_full_rewrite ? torch::empty_like(gseg) : gseg.clone();
  auto empty_i64 = torch::zeros({K > 0 ? K : 1}, torch::kInt64);
  if (K == 0)
    return {ridx_out, gseg_out, empty_i64, empty_i64, empty_i64,
            torch::zeros({1}, torch::kInt64)};
  auto starts_cpu = starts.to(torch::kCPU).to(torch::kInt64);
  auto counts_cpu = counts.to(torch::kCPU).to(torch::kInt64);
  int64_t total_chunks = 0;
  auto chunk_off_cpu = torch::zeros({K + 1}, torch::kInt64);
  {
    auto acc = chunk_off_cpu.accessor<int64_t, 1>();
    auto cacc = counts_cpu.accessor<int64_t, 1>();
    for (int k = 0; k < K; ++k)
      acc[k + 1] = acc[k] + (cacc[k] + PART_CHUNK - 1) / PART_CHUNK;
    total_chunks = acc[K];
  }
  if (total_chunks == 0)
    return {ridx_out, gseg_out, empty_i64, empty_i64, empty_i64,
            torch::zeros({1}, torch::kInt64)};
  auto stream0 = c10::hip::getCurrentHIPStream();
  static thread_local PinnedStager part_meta_stager;
  auto meta_cpu = part_meta_stager.get(6 * K + 1);
  cat_into_pinned(meta_cpu, {starts_cpu, counts_cpu, chunk_off_cpu,
                             split_feat.to(torch::kCPU).to(torch::kInt64),
                             split_bin.to(torch::kCPU).to(torch::kInt64),
                             default_left.to(torch::kCPU).to(torch::kInt64)});
  auto meta = meta_cpu.to(dev, /*non_blocking=*/true);
  part_meta_stager.mark(stream0.stream());
  int64_t* mp = meta.data_ptr<int64_t>();

  auto stream = c10::hip::getCurrentHIPStream();
  auto block_counts = torch::empty({total_chunks},
      torch::TensorOptions().dtype(torch::kInt32).device(dev));
  auto flags = torch::empty({ridx.size(0)},
      torch::TensorOptions().dtype(torch::kUInt8).device(dev));
  hipLaunchKernelGGL(partition_count_kernel, dim3((uint32_t)total_chunks),
                     dim3(PART_THREADS), 0, stream.stream(),
                     bins.data_ptr<uint8_t>(), ridx.data_ptr<int32_t>(),
                     mp, (const int64_t*)nullptr,
                     block_counts.data_ptr<int32_t>(),
                     flags.data_ptr<uint8_t>(), K, bins.stride(0),
                     bins_t.numel() ? bins_t.data_ptr<uint8_t>() : nullptr,
                     bins.size(0));
  auto left_before = torch::empty({total_chunks},
      torch::TensorOptions().dtype(torch::kInt64).device(dev));
  auto node_left_total = torch::empty({K},
      torch::TensorOptions().dtype(torch::kInt64).device(dev));
  hipLaunchKernelGGL(partition_prefix_kernel, dim3((uint32_t)K), dim3(256),
                     0, stream0.stream(),
                     block_counts.data_ptr<int32_t>(), mp,
                     (const int64_t*)nullptr,
                     left_before.data_ptr<int64_t>(),
                     node_left_total.data_ptr<int64_t>(), K);
  auto tc = torch::full({1}, total_chunks, torch::kInt64);
  return {ridx_out, gseg_out, meta, left_before, node_left_total, tc,
          flags};
}

std::vector<torch::Tensor> partition_rows_finish(
    torch::Tensor ridx, torch::Tensor ridx_out, torch::Tensor gseg,
    torch::Tensor gseg_out, torch::Tensor meta, torch::Tensor left_before,
    torch::Tensor node_left_total, torch::Tensor tc_t,
    torch::Tensor flags) {
  const int K = (int)node_left_total.size(0);
  const int64_t total_chunks = tc_t.item<int64_t>();
  if (total_chunks == 0)
    return {ridx_out, torch::zeros({K}, torch::kInt64), gseg_out};
  auto stream = c10::hip::getCurrentHIPStream();
  static thread_local PinnedStager nl_stager;
  auto node_left_total_cpu_pin = nl_stager.get(K);
  node_left_total_cpu_pin.copy_(node_left_total);  // sync: host needs it
  auto node_left_total_cpu = node_left_total_cpu_pin.clone();
  int64_t* mp = meta.data_ptr<int64_t>();
  hipLaunchKernelGGL(partition_scatter_kernel, dim3((uint32_t)total_chunks),
                     dim3(PART_THREADS), 0, stream.stream(),
                     flags.data_ptr<uint8_t>(), ridx.data_ptr<int32_t>(),
                     ridx_out.data_ptr<int32_t>(),
                     gseg.numel() ? (const int2*)gseg.data_ptr<int32_t>()
                                  : nullptr,
                     gseg_out.numel() ? (int2*)gseg_out.data_ptr<int32_t>()
                                      : nullptr,
                     mp, (const int64_t*)nullptr,
                     left_before.data_ptr<int64_t>(),
                     node_left_total.data_ptr<int64_t>(), K);
  return {ridx_out, node_left_total_cpu, gseg_out};
}

std::vector<torch::Tensor> partition_rows(torch::Tensor bins, torch::Tensor ridx,
                                          torch::Tensor starts, torch::Tensor counts,
                                          torch::Tensor split_feat,
                                          torch::Tensor split_bin,
                                          torch::Tensor default_left,
                                          torch::Tensor gseg,
                                          torch::Tensor bins_t) {
  auto st = partition_rows_begin(bins, ridx, starts, counts, split_feat,
                                 split_bin, default_left, gseg, bins_t,
                                 /*gseg_full_rewrite=*/false);
  if (st.size() == 6) {  // K==0 / no chunks: nothing to scatter
    return {st[0], torch::zeros({(int64_t)starts.size(0)}, torch::kInt64),
            st[1]};
  }
  return partition_rows_finish(ridx, st[0], gseg, st[1], st[2], st[3],
                               st[4], st[5], st[6]);
}


// ---------------------------------------------------------------------------
// predict_trees_lds: tree-tiled serving walk. Trees are packed into
// contiguous-node TILES (<= PRED_TILE_NODES nodes, 64 KiB) that a block
// stages into LDS once per 2048-row chunk; every node access during the
// walk is then an LDS read instead of a random L1/L2 hit - the plain
// kernel is node-fetch latency bound. Rows' feature reads stay in L1
// (each row re-reads its own 1-2 lines). Falls back to the plain kernel
// for single trees deeper than the tile (host decides).
// ---------------------------------------------------------------------------
#define PRED_TILE_NODES 4096
#define PRED_ROWS_PER_THREAD 8
__global__ __launch_bounds__(256) void predict_trees_lds_kernel(
    const float* __restrict__ X, const uint4* __restrict__ nodes,
    const int32_t* __restrict__ tree_ptr,
    const int32_t* __restrict__ tile_ptr,  // [n_tiles+1] tree indices
    float* __restrict__ out, float tree_weight, int64_t n, int F, int T,
    int n_tiles) {
  __shared__ uint4 lds_nodes[PRED_TILE_NODES];
  const int64_t rows_per_chunk = (int64_t)blockDim.x * PRED_ROWS_PER_THREAD;
  const int64_t n_chunks = (n + rows_per_chunk - 1) / rows_per_chunk;
  for (int64_t chunk = blockIdx.x; chunk < n_chunks; chunk += gridDim.x) {
    const int64_t row0 = chunk * rows_per_chunk + threadIdx.x;
    float acc[PRED_ROWS_PER_THREAD];
    #pragma unroll
    for (int r = 0; r < PRED_ROWS_PER_THREAD; ++r) acc[r] = 0.0f;
    for (int tile = 0; tile < n_tiles; ++tile) {
      const int t0 = tile_ptr[tile], t1 = tile_ptr[tile + 1];
      const int nb0 = tree_ptr[t0];
      const int span = tree_ptr[t1] - nb0;
      __syncthreads();  // previous walk done before overwriting LDS
      for (int i = threadIdx.x; i < span; i += blockDim.x)
        lds_nodes[i] = nodes[nb0 + i];
      __syncthreads();
      for (int t = t0; t < t1; ++t) {
        const int rel = tree_ptr[t] - nb0;
        #pragma unroll
        for (int r = 0; r < PRED_ROWS_PER_THREAD; ++r) {
          const int64_t row = row0 + (int64_t)r * blockDim.x;
          if (row >= n) continue;
          uint4 p = lds_nodes[rel];
          while (!(p.x & 0x80000000u)) {
            const float v = X[row * (int64_t)F + (p.x & 0x3FFFFFFFu)];
            const bool goleft =
                isnan(v) ? ((p.x & 0x40000000u) != 0)
                         : (v < __uint_as_float(p.y));
            p = lds_nodes[rel + (int)p.z + (goleft ? 0 : 1)];
          }
          acc[r] += __uint_as_float(p.y);
        }
      }
    }
    #pragma unroll
    for (int r = 0; r < PRED_ROWS_PER_THREAD; ++r) {
      const int64_t row = row0 + (int64_t)r * blockDim.x;
      if (row < n) out[row] += acc[r] * tree_weight;
    }
  }
}

void predict_trees(torch::Tensor X, torch::Tensor feat, torch::Tensor thr,
                   torch::Tensor left, torch::Tensor default_left,
                   torch::Tensor value, torch::Tensor tree_ptr,
                   torch::Tensor out, double tree_weight) {
  int64_t n = X.size(0);
  int F = (int)X.size(1);
  int T = (int)tree_ptr.size(0) - 1;
  if (n == 0 || T <= 0) return;
  auto stream = c10::hip::getCurrentHIPStream();
  int64_t n_nodes = feat.size(0);
  auto packed = torch::empty({n_nodes, 4},
      torch::TensorOptions().dtype(torch::kInt32).device(X.device()));
  hipLaunchKernelGGL(pack_nodes_kernel, dim3((uint32_t)ceil_div(n_nodes, 256)),
                     dim3(256), 0, stream.stream(),
                     feat.data_ptr<int32_t>(), thr.data_ptr<float>(),
                     left.data_ptr<int32_t>(), default_left.data_ptr<uint8_t>(),
                     value.data_ptr<float>(),
                     (uint4*)packed.data_ptr<int32_t>(), n_nodes);
  // tree-tiled LDS path: pack trees into <= PRED_TILE_NODES-node tiles
  // (CPU-side scan over tree_ptr; falls back to the plain kernel when a
  // single tree exceeds the tile or when disabled)
  bool use_lds = true;
  if (const char* e = getenv("RXGB_PREDICT_LDS")) use_lds = atoi(e) != 0;
  std::vector<int32_t> tiles;
  if (use_lds) {
    auto tp_cpu = tree_ptr.to(torch::kCPU).to(torch::kInt32);
    auto tp = tp_cpu.accessor<int32_t, 1>();
    tiles.push_back(0);
    int t = 0;
    while (t < T) {
      int t_end = t;
      while (t_end < T && tp[t_end + 1] - tp[t] <= PRED_TILE_NODES)
        ++t_end;
      if (t_end == t) { use_lds = false; break; }  // one tree > tile
      tiles.push_back(t_end);
      t = t_end;
    }
  }
  if (use_lds) {
    auto tile_t = torch::from_blob(
        tiles.data(), {(int64_t)tiles.size()},
        torch::TensorOptions().dtype(torch::kInt32)).clone().to(X.device());
    const int64_t rows_per_chunk = 256 * PRED_ROWS_PER_THREAD;
    int64_t blocks = std::min<int64_t>(
        ceil_div(n, rows_per_chunk), 8192);
    hipLaunchKernelGGL(predict_trees_lds_kernel, dim3((uint32_t)blocks),
                       dim3(256), 0, stream.stream(),
                       X.data_ptr<float>(),
                       (const uint4*)packed.data_ptr<int32_t>(),
                       tree_ptr.data_ptr<int32_t>(),
                       tile_t.data_ptr<int32_t>(),
                       out.data_ptr<float>(), (float)tree_weight, n, F, T,
                       (int)tiles.size() - 1);
    return;
  }
  // measured: R=4 ILP loses to plain TLP here (75.8 vs 94.5 M rows/s on
  // 100 trees x depth 8) - one row per thread with a full grid wins
  int64_t blocks = std::min<int64_t>(ceil_div(n, 256), 8192);
  hipLaunchKernelGGL((predict_trees_kernel<1>), dim3(blocks), dim3(256), 0,
                     stream.stream(), X.data_ptr<float>(),
                     (const uint4*)packed.data_ptr<int32_t>(),
                     tree_ptr.data_ptr<int32_t>(),
                     out.data_ptr<float>(), (float)tree_weight, n, F, T);
}

void update_margins(torch::Tensor margin, torch::Tensor ridx,
                    torch::Tensor starts, torch::Tensor counts,
                    torch::Tensor leaf_vals, torch::Tensor ridx_b,
                    torch::Tensor parity) {
  const int K = (int)starts.size(0);
  if (K == 0) return;
  auto dev = margin.device();
  auto starts_cpu = starts.to(torch::kCPU).to(torch::kInt64);
  auto counts_cpu = counts.to(torch::kCPU).to(torch::kInt64);
  auto chunk_off_cpu = torch::zeros({K + 1}, torch::kInt64);
  int64_t total_chunks = 0;
  {
    auto acc = chunk_off_cpu.accessor<int64_t, 1>();
    auto cacc = counts_cpu.accessor<int64_t, 1>();
    for (int k = 0; k < K; ++k)
      acc[k + 1] = acc[k] + (cacc[k] + MARGIN_CHUNK - 1) / MARGIN_CHUNK;
    total_chunks = acc[K];
  }
  if (total_chunks == 0) return;
  auto stream0 = c10::hip::getCurrentHIPStream();
  static thread_local PinnedStager margin_meta_stager;
  const bool have_parity = parity.numel() > 0;
  auto meta_cpu = margin_meta_stager.get((have_parity ? 4 : 3) * K + 1);
  if (have_parity) {
    cat_into_pinned(meta_cpu, {starts_cpu, counts_cpu, chunk_off_cpu,
                               parity.to(torch::kCPU).to(torch::kInt64)});
  } else {
    cat_into_pinned(meta_cpu, {starts_cpu, counts_cpu, chunk_off_cpu});
  }
  auto meta = meta_cpu.to(dev, /*non_blocking=*/true);
  margin_meta_stager.mark(stream0.stream());
  int64_t* mp = meta.data_ptr<int64_t>();
  // leaf values ride their own pinned stager (fp32; the int64 meta
  // stager cannot carry them without a bitcast)
  static thread_local PinnedStager margin_lv_stager;
  auto lv_cpu = margin_lv_stager.get(K, torch::kFloat32);
  lv_cpu.copy_(leaf_vals.to(torch::kFloat32));
  auto lv = lv_cpu.to(dev, /*non_blocking=*/true);
  margin_lv_stager.mark(stream0.stream());
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(update_margins_kernel, dim3((uint32_t)total_chunks),
                     dim3(PART_THREADS), 0, stream.stream(),
                     margin.data_ptr<float>(), ridx.data_ptr<int32_t>(),
                     ridx_b.numel() ? ridx_b.data_ptr<int32_t>()
                                    : ridx.data_ptr<int32_t>(),
                     mp, mp + 2 * K,
                     have_parity ? mp + 3 * K + 1 : nullptr,
                     lv.data_ptr<float>(), K);
}

torch::Tensor lambdarank_grad(torch::Tensor margin, torch::Tensor label,
                              torch::Tensor group_ptr, torch::Tensor rank,
                              torch::Tensor idcg, bool use_ndcg) {
  const int64_t n = margin.size(0);
  const int G = (int)group_ptr.size(0) - 1;
  auto out = torch::zeros({n, 2},
      torch::TensorOptions().dtype(torch::kFloat32).device(margin.device()));
  if (n == 0 || G <= 0) return out;
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(lambdarank_kernel, dim3(G), dim3(256), 0,
                     stream.stream(), margin.data_ptr<float>(),
                     label.data_ptr<float>(), group_ptr.data_ptr<int64_t>(),
                     rank.data_ptr<int32_t>(),
                     use_ndcg ? idcg.data_ptr<double>() : nullptr,
                     (float2*)out.data_ptr<float>(), use_ndcg ? 1 : 0);
  return out;
}


// ---------------------------------------------------------------------------
// grad_fused: objective gradient + hessian + |g|/|h| max in ONE pass.
// Replaces the 5-kernel torch chain (sigmoid, sub, mul, clamp, stack)
// plus the two abs().max() reduction passes for the hot objectives -
// the margin/label are read once and the fp32 gpair written once.
// The f32 op ORDER replicates torch's eager kernels exactly
// (p = 1/(1+expf(-x)); h = fmaxf(p*(1-p), eps); per-factor multiplies in
// the same sequence), so models stay bitwise-identical to the CPU
// oracle. The max reduction is order-free (fmax is associative).
// MODE 0: reg:squarederror (g = m - y, h = 1)
// MODE 1: binary:logistic  (g = p - y, h = max(p(1-p), 1e-16))
// ---------------------------------------------------------------------------
template <int MODE>
__global__ __launch_bounds__(256) void grad_fused_kernel(
    const float* __restrict__ margin, const float* __restrict__ label,
    const float* __restrict__ weight,  // nullable
    float spw, float2* __restrict__ gpair,
    float* __restrict__ block_max,  // [gridDim.x, 2] per-block maxes
    int64_t n) {
  float gmax = 0.0f, hmax = 0.0f;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const float x = margin[i];
    const float y = label[i];
    float g, h;
    if constexpr (MODE == 1) {
      const float p = 1.0f / (1.0f + expf(-x));
      g = p - y;
      h = fmaxf(p * (1.0f - p), 1e-16f);
      if (spw != 1.0f) {
        const float w = 1.0f + (spw - 1.0f) * y;
        g *= w;
        h *= w;
      }
    } else {
      g = x - y;
      h = 1.0f;
    }
    if (weight != nullptr) {
      const float w = weight[i];
      g *= w;
      h *= w;
    }
    gpair[i] = {g, h};
    gmax = fmaxf(gmax, fabsf(g));
    hmax = fmaxf(hmax, fabsf(h));
  }
  // wave reduce -> LDS -> one [2]-float store per BLOCK. A global
  // atomicMax per wave serialized 43k atomics on TWO L2 addresses and
  // cost ~10x the kernel's memory traffic (measured 496 us for 186 MB);
  // per-block maxes + a tiny torch amax reduction have zero contention.
  __shared__ float red[2 * 256 / WAVE];
  #pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    gmax = fmaxf(gmax, __shfl_down(gmax, off, WAVE));
    hmax = fmaxf(hmax, __shfl_down(hmax, off, WAVE));
  }
  const int wid = threadIdx.x / WAVE;
  if ((threadIdx.x & (WAVE - 1)) == 0) {
    red[2 * wid] = gmax;
    red[2 * wid + 1] = hmax;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    const int nw = blockDim.x / WAVE;
    float g = red[0], h = red[1];
    for (int w = 1; w < nw; ++w) {
      g = fmaxf(g, red[2 * w]);
      h = fmaxf(h, red[2 * w + 1]);
    }
    block_max[2 * (size_t)blockIdx.x] = g;
    block_max[2 * (size_t)blockIdx.x + 1] = h;
  }
}

std::vector<torch::Tensor> grad_fused(torch::Tensor margin,
                                      torch::Tensor label,
                                      torch::Tensor weight, double spw,
                                      int64_t mode) {
  TORCH_CHECK(on_gpu(margin) && margin.dtype() == torch::kFloat32);
  const int64_t n = margin.numel();
  auto dev = margin.device();
  auto gpair = torch::empty(
      {n, 2}, torch::TensorOptions().dtype(torch::kFloat32).device(dev));
  if (n == 0) {
    return {gpair, torch::zeros({2}, torch::TensorOptions()
                                         .dtype(torch::kFloat32)
                                         .device(dev))};
  }
  auto stream = c10::hip::getCurrentHIPStream();
  const int64_t blocks = std::min<int64_t>(ceil_div(n, 256 * 16), 4096);
  auto block_max = torch::empty(
      {blocks, 2},
      torch::TensorOptions().dtype(torch::kFloat32).device(dev));
  const float* wp =
      weight.numel() ? weight.data_ptr<float>() : nullptr;
  if (mode == 1) {
    hipLaunchKernelGGL((grad_fused_kernel<1>), dim3((uint32_t)blocks),
                       dim3(256), 0, stream.stream(),
                       margin.data_ptr<float>(), label.data_ptr<float>(),
                       wp, (float)spw,
                       (float2*)gpair.data_ptr<float>(),
                       block_max.data_ptr<float>(), n);
  } else {
    hipLaunchKernelGGL((grad_fused_kernel<0>), dim3((uint32_t)blocks),
                       dim3(256), 0, stream.stream(),
                       margin.data_ptr<float>(), label.data_ptr<float>(),
                       wp, (float)spw,
                       (float2*)gpair.data_ptr<float>(),
                       block_max.data_ptr<float>(), n);
  }
  // fmax is order-free, so this equals abs().max() bitwise
  auto absmax = std::get<0>(block_max.max(0));
  return {gpair, absmax};
}


// ---------------------------------------------------------------------------
// Fused eval metrics: ONE pass over [margin, label(, weight)] replaces
// the 4-6 torch f64 elementwise/reduction kernels per metric per round.
// logloss: per-block (sum w*ll, sum w) f64 partials -> torch .sum(0)
// (fixed-order reduction, deterministic). AUC: unweighted int64
// pos/neg score-bin histograms via global atomics (exact).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void eval_logloss_kernel(
    const float* __restrict__ margin, const float* __restrict__ label,
    const float* __restrict__ weight,  // nullable
    double* __restrict__ block_out,    // [gridDim.x, 2]
    int64_t n) {
  double ll = 0.0, wsum = 0.0;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    // same math as metrics.LogLoss: sigmoid in f64, clamped p
    double p = 1.0 / (1.0 + exp(-(double)margin[i]));
    p = fmin(fmax(p, 1e-16), 1.0 - 1e-16);
    const double y = (double)label[i];
    const double w = weight ? (double)weight[i] : 1.0;
    ll += w * -(y * log(p) + (1.0 - y) * log(1.0 - p));
    wsum += w;
  }
  __shared__ double red[2 * 256 / WAVE];
  #pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) {
    ll += __shfl_down(ll, off, WAVE);
    wsum += __shfl_down(wsum, off, WAVE);
  }
  const int wid = threadIdx.x / WAVE;
  if ((threadIdx.x & (WAVE - 1)) == 0) {
    red[2 * wid] = ll;
    red[2 * wid + 1] = wsum;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    const int nw = blockDim.x / WAVE;
    double a = red[0], b = red[1];
    for (int w = 1; w < nw; ++w) {
      a += red[2 * w];
      b += red[2 * w + 1];
    }
    block_out[2 * (size_t)blockIdx.x] = a;
    block_out[2 * (size_t)blockIdx.x + 1] = b;
  }
}

torch::Tensor eval_logloss(torch::Tensor margin, torch::Tensor label,
                           torch::Tensor weight) {
  const int64_t n = margin.numel();
  auto dev = margin.device();
  const int64_t blocks = std::min<int64_t>(ceil_div(n, 256 * 16), 2048);
  auto block_out = torch::empty(
      {std::max<int64_t>(blocks, 1), 2},
      torch::TensorOptions().dtype(torch::kFloat64).device(dev));
  if (n == 0) return torch::zeros({2}, block_out.options());
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(eval_logloss_kernel, dim3((uint32_t)blocks),
                     dim3(256), 0, stream.stream(),
                     margin.data_ptr<float>(), label.data_ptr<float>(),
                     weight.numel() ? weight.data_ptr<float>() : nullptr,
                     block_out.data_ptr<double>(), n);
  return block_out.sum(0);  // fixed-order torch reduction
}

__global__ __launch_bounds__(256) void eval_auc_hist_kernel(
    const float* __restrict__ margin, const float* __restrict__ label,
    long long* __restrict__ hist,  // [2*B]: neg plane | pos plane
    int B, int64_t n) {
  // Per-WG LDS u32 histogram (2*B*4 = 128 KiB dynamic LDS), merged
  // once with skip-if-zero global atomics. The direct-global version
  // serialized on the few bins a saturated sigmoid concentrates mass
  // into: measured 850 us/round on HIGGS (~30x this kernel's memory
  // cost). Counts per WG stay far below 2^32.
  extern __shared__ unsigned int lds_auc[];
  for (int i = threadIdx.x; i < 2 * B; i += blockDim.x) lds_auc[i] = 0u;
  __syncthreads();
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const double p = 1.0 / (1.0 + exp(-(double)margin[i]));
    long long b = (long long)(p * (double)B);  // trunc, matches .long()
    if (b > B - 1) b = B - 1;
    if (b < 0) b = 0;
    const int pos = label[i] > 0.5f ? 1 : 0;
    atomicAdd(&lds_auc[(size_t)pos * B + b], 1u);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < 2 * B; i += blockDim.x) {
    const unsigned int v = lds_auc[i];
    if (v)
      atomicAdd((unsigned long long*)&hist[i], (unsigned long long)v);
  }
}

torch::Tensor eval_auc_hist(torch::Tensor margin, torch::Tensor label,
                            int64_t n_bins) {
  const int64_t n = margin.numel();
  auto dev = margin.device();
  auto hist = torch::zeros(
      {2 * n_bins},
      torch::TensorOptions().dtype(torch::kInt64).device(dev));
  if (n == 0) return hist;
  auto stream = c10::hip::getCurrentHIPStream();
  const size_t lds = 2 * (size_t)n_bins * sizeof(unsigned int);
  static bool lds_attr_set = false;
  if (!lds_attr_set) {
    // dynamic LDS above 64 KiB needs the opt-in attribute (160 KiB/CU)
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&eval_auc_hist_kernel),
        hipFuncAttributeMaxDynamicSharedMemorySize, 160 * 1024);
    lds_attr_set = true;
  }
  const int64_t blocks = std::min<int64_t>(ceil_div(n, 256 * 64), 256);
  hipLaunchKernelGGL(eval_auc_hist_kernel, dim3((uint32_t)blocks),
                     dim3(256), lds, stream.stream(),
                     margin.data_ptr<float>(), label.data_ptr<float>(),
                     reinterpret_cast<long long*>(hist.data_ptr<int64_t>()),
                     (int)n_bins, n);
  return hist;  // [neg | pos] planes
}


// ---------------------------------------------------------------------------
// depth_step: one C++ call runs a whole depth of the fused single-sync
// loop - stage control data, zero+build the histogram, derive siblings,
// scan, plan+partition, enqueue the pull and record the pull event,
// then launch the scatter. The python loop's ~30 torch/pybind dispatches
// per depth (~100 us of host time each depth) collapse to one call +
// one wait + numpy bookkeeping. Single-GPU only: the distributed path
// keeps the python structure (it interleaves the RCCL allreduce).
// ---------------------------------------------------------------------------
static hipEvent_t g_pull_event = nullptr;

void wait_pull_event() {
  if (g_pull_event) (void)hipEventSynchronize(g_pull_event);
}

std::vector<torch::Tensor> depth_step(
    torch::Tensor bins, torch::Tensor bins_t, torch::Tensor gseg,
    torch::Tensor ridx, torch::Tensor ridx_dest, torch::Tensor prev_hist,
    torch::Tensor hist, torch::Tensor depth_meta_cpu, int64_t KK,
    int64_t nd, int64_t K_built, torch::Tensor starts_cpu,
    torch::Tensor counts_cpu, torch::Tensor fb, double scale_g,
    double scale_h, double lam, double alpha, double gamma, double mcw,
    torch::Tensor mono, torch::Tensor bounds, torch::Tensor allowed,
    int64_t n_bins, int64_t chunk_bound) {
  auto dev = bins.device();
  auto stream = c10::hip::getCurrentHIPStream();
  const int F = (int)bins.size(1);

  // ONE pinned H2D for all of this depth's control data
  static thread_local PinnedStager depth_meta_stager;
  auto meta_cpu_pin = depth_meta_stager.get(depth_meta_cpu.numel());
  meta_cpu_pin.copy_(depth_meta_cpu);
  auto dmeta = meta_cpu_pin.to(dev, /*non_blocking=*/true);
  depth_meta_stager.mark(stream.stream());
  // layout: [sumg KK | sumh KK | starts KK | counts KK | pslot nd | sib nd]
  auto pg = dmeta.narrow(0, 0, KK);
  auto ph = dmeta.narrow(0, KK, KK);
  auto d_starts = dmeta.narrow(0, 2 * KK, KK);
  auto d_counts = dmeta.narrow(0, 3 * KK, KK);

  // built block: zero + histogram
  hist.narrow(0, 0, K_built).zero_();
  build_histogram(bins, gseg, ridx, starts_cpu, counts_cpu, n_bins, 0, F,
                  hist, /*pregathered=*/true);
  // sibling = parent - built
  if (nd > 0) {
    auto pslots = dmeta.narrow(0, 4 * KK, nd);
    auto spos = dmeta.narrow(0, 4 * KK + nd, nd);
    auto derived = hist.narrow(0, K_built, nd);
    torch::sub_out(derived, prev_hist.index_select(0, pslots),
                   hist.index_select(0, spos));
  }
  // scan
  auto fs = find_splits(hist, pg, ph, fb, scale_g, scale_h, lam, alpha,
                        gamma, mcw, mono, bounds, allowed,
                        /*pull=*/false);
  auto packed = fs[0];
  // device-planned partition (writes into ridx_dest, no clones)
  auto st = partition_rows_from_packed(bins, ridx, d_starts, d_counts,
                                       packed, gseg, bins_t, chunk_bound,
                                       ridx_dest);
  // record the pull event BETWEEN the pull copies and the scatter
  if (!g_pull_event)
    (void)hipEventCreateWithFlags(&g_pull_event, hipEventDisableTiming);
  (void)hipEventRecord(g_pull_event, stream.stream());
  partition_scatter_from_packed(ridx, st[0], gseg, st[1], st[3], st[4],
                                st[5], st[6], st[7], st[8]);
  return {st[0], st[1], st[2]};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("quantize_gpair", &quantize_gpair, "quantize gradient pairs");
  m.def("grad_fused", &grad_fused, "fused objective gradient + absmax");
  m.def("eval_logloss", &eval_logloss, "fused logloss eval");
  m.def("eval_auc_hist", &eval_auc_hist, "fused AUC score histograms");
  m.def("bin_matrix", &bin_matrix, "bin feature matrix");
  m.def("build_histogram", &build_histogram, "build gradient histograms");
  m.def("find_splits", &find_splits, "best-split scan");
  m.def("partition_rows", &partition_rows, "stable row partition");
  m.def("partition_rows_begin", &partition_rows_begin,
        "count+prefix phase (returns before the host sync)");
  m.def("partition_rows_finish", &partition_rows_finish,
        "left-totals pull + scatter phase");
  m.def("partition_rows_from_packed", &partition_rows_from_packed,
        "device-planned partition consuming find_splits packed output");
  m.def("partition_scatter_from_packed", &partition_scatter_from_packed,
        "scatter phase of the device-planned partition");
  m.def("depth_step", &depth_step,
        "one fused single-sync depth (single-GPU fast path)");
  m.def("wait_pull_event", &wait_pull_event,
        "host wait for the depth_step pull event");
  m.def("predict_trees", &predict_trees, "tree-walk prediction");
  m.def("update_margins", &update_margins, "leaf margin update");
  m.def("lambdarank_grad", &lambdarank_grad, "pairwise lambdarank gradients");
}
