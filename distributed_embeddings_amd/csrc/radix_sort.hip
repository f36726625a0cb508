// Hand-written LSD radix sort, tuned for CDNA4 wave64 — replaces rocPRIM in
// the sparse-backward pipeline ("hand-written LDS-staged
// segmented sort ... with rocPRIM only as a bring-up crutch").
//
// The pipeline packs (id << 32 | position) into ONE u64 key, so the sort
// moves a single array; digits start at bit 32 (custom_radix_sort_keys,
// begin_bit/end_bit args).  8-bit digits, one pass per digit:
//   1. rs_histogram_k: per-block bucket counts (LDS), written digit-major
//      [bucket][block] for the stable global scan;
//   2. exclusive scan over the [256 x nblocks] table (hand-written:
//      per-chunk partials + single-wave top scan + add-back);
//   3. rs_scatter_k: WAVES waves per block, each owning a consecutive
//      1/WAVES slice of the tile with a PRIVATE running-offset row
//      (barrier-free scatter loop); within each 64-element group the stable
//      rank comes from 8 __ballot rounds (bucket-bit match masks) + popcount
//      over the preceding-lane mask.  Per-wave bases fold in earlier slices'
//      counts so the sort stays stable.
//
// Tile geometry is a (WAVES, IPT) template: at DLRM-backward sizes
// (~200k keys) the original 4x16 = 4096-element tile yields only ~52
// blocks — far below what 256 CUs need — so smaller tiles win despite the
// larger histogram/scan table.  The variant is chosen at dispatch
// (DE_SORT_VARIANT env overrides for measurement; see tools/bench_sort3.py).
// Keys' digit bits must be non-negative ids (the pipeline masks OOB ids to
// the `vocab` sentinel before packing).

#include <hip/hip_runtime.h>

#include <cstdlib>

#include "ops_api.h"

#define WAVE 64
#define RS_RADIX 256

static inline int64_t rs_cdiv(int64_t a, int64_t b) { return (a + b - 1) / b; }

// two-level exclusive scan over m = 256*nblocks entries
__global__ void rs_scan_partials(const int32_t* __restrict__ in, int64_t m,
                                 int32_t* __restrict__ block_sums,
                                 int32_t* __restrict__ out, int chunk) {
  // each block scans its contiguous chunk serially with one wave
  __shared__ int32_t carry;
  const int lane = threadIdx.x;
  const int64_t start = (int64_t)blockIdx.x * chunk;
  const int64_t end = min(start + (int64_t)chunk, m);
  if (lane == 0) carry = 0;
  __syncthreads();
  for (int64_t base = start; base < end; base += WAVE) {
    const int64_t i = base + lane;
    const int32_t v = i < end ? in[i] : 0;
    // wave-inclusive scan via shuffles
    int32_t s = v;
    for (int off = 1; off < WAVE; off <<= 1) {
      const int32_t o = __shfl_up(s, off);
      if (lane >= off) s += o;
    }
    if (i < end) out[i] = s - v + carry;  // exclusive
    __syncthreads();
    if (lane == WAVE - 1) carry += s;
    __syncthreads();
  }
  if (lane == 0) block_sums[blockIdx.x] = carry;
}

__global__ void rs_scan_top(int32_t* __restrict__ block_sums, int nb) {
  // single wave: exclusive scan of block sums (nb <= a few thousand)
  __shared__ int32_t carry;
  const int lane = threadIdx.x;
  if (lane == 0) carry = 0;
  __syncthreads();
  for (int base = 0; base < nb; base += WAVE) {
    const int i = base + lane;
    const int32_t v = i < nb ? block_sums[i] : 0;
    int32_t s = v;
    for (int off = 1; off < WAVE; off <<= 1) {
      const int32_t o = __shfl_up(s, off);
      if (lane >= off) s += o;
    }
    if (i < nb) block_sums[i] = s - v + carry;
    __syncthreads();
    if (lane == WAVE - 1) carry += s;
    __syncthreads();
  }
}

__global__ void rs_scan_addback(int32_t* __restrict__ out, int64_t m,
                                const int32_t* __restrict__ block_sums,
                                int chunk) {
  const int64_t start = (int64_t)blockIdx.x * chunk;
  const int64_t end = min(start + (int64_t)chunk, m);
  const int32_t add = block_sums[blockIdx.x];
  for (int64_t i = start + threadIdx.x; i < end; i += blockDim.x) {
    out[i] += add;
  }
}


// Keys-only variant: the backward pipeline packs (id << 32 | position) into
// one u64, so the sort moves a single array (half the scatter write
// traffic of pairs) and digits start at bit 32.
template <int WAVES, int IPT>
__global__ void rs_histogram_k(const uint64_t* __restrict__ keys, int64_t n,
                               int shift, int32_t* __restrict__ hist,
                               int64_t nblocks) {
  constexpr int TILE = WAVES * WAVE * IPT;
  __shared__ int32_t lhist[RS_RADIX];
  const int t = threadIdx.x;
  for (int b = t; b < RS_RADIX; b += WAVES * WAVE) lhist[b] = 0;
  __syncthreads();
  const int64_t start = (int64_t)blockIdx.x * TILE;
  const int64_t end = min(start + (int64_t)TILE, n);
  for (int64_t i = start + t; i < end; i += WAVES * WAVE) {
    const int b = (int)((keys[i] >> shift) & 0xff);
    atomicAdd(&lhist[b], 1);
  }
  __syncthreads();
  for (int b = t; b < RS_RADIX; b += WAVES * WAVE)
    hist[(int64_t)b * nblocks + blockIdx.x] = lhist[b];
}

// WAVES waves per block; each wave owns a consecutive slice of the tile with
// a PRIVATE running-offset row (no barriers in the scatter loop).
// Stability: per-wave bases include the counts of earlier slices.
template <int WAVES, int IPT>
__global__ void rs_scatter_k(const uint64_t* __restrict__ keys_in, int64_t n,
                             int shift, const int32_t* __restrict__ offsets,
                             int64_t nblocks, uint64_t* __restrict__ keys_out) {
  constexpr int TILE = WAVES * WAVE * IPT;
  __shared__ int32_t cnt[WAVES][RS_RADIX];
  __shared__ int32_t run[WAVES][RS_RADIX];
  const int t = threadIdx.x;
  const int wave = t >> 6;
  const int lane = t & (WAVE - 1);
  for (int b = t; b < RS_RADIX; b += WAVES * WAVE) {
#pragma unroll
    for (int w = 0; w < WAVES; ++w) cnt[w][b] = 0;
  }
  __syncthreads();
  const int64_t start = (int64_t)blockIdx.x * TILE;
  const int64_t end = min(start + (int64_t)TILE, n);
  const int64_t qs = min(start + (int64_t)wave * WAVE * IPT, end);
  const int64_t qe = min(qs + (int64_t)WAVE * IPT, end);
  // phase 1: per-slice bucket counts
  for (int64_t i = qs + lane; i < qe; i += WAVE) {
    const int b = (int)((keys_in[i] >> shift) & 0xff);
    atomicAdd(&cnt[wave][b], 1);
  }
  __syncthreads();
  // phase 2: per-wave bases = global offset + earlier slices' counts
  for (int b = t; b < RS_RADIX; b += WAVES * WAVE) {
    int base = offsets[(int64_t)b * nblocks + blockIdx.x];
#pragma unroll
    for (int w = 0; w < WAVES; ++w) {
      run[w][b] = base;
      base += cnt[w][b];
    }
  }
  __syncthreads();
  // phase 3: each wave scatters its slice, 64-element groups in order
  for (int64_t base = qs; base < qe; base += WAVE) {
    const int64_t i = base + lane;
    const bool active = i < qe;
    const uint64_t key = active ? keys_in[i] : 0;
    const int b = (int)((key >> shift) & 0xff);
    uint64_t same = ~0ull;
    #pragma unroll
    for (int bit = 0; bit < 8; ++bit) {
      const uint64_t setmask = __ballot((b >> bit) & 1);
      same &= ((b >> bit) & 1) ? setmask : ~setmask;
    }
    const uint64_t act = __ballot(active);
    same &= act;
    const uint64_t before = same & ((1ull << lane) - 1ull);
    const int rank = __popcll(before);
    const int leader = __ffsll((unsigned long long)same) - 1;
    int base_off = 0;
    if (active && lane == leader) {
      base_off = run[wave][b];
      run[wave][b] += __popcll(same);
    }
    if (active) {
      const int bo = __shfl(base_off, leader);
      keys_out[bo + rank] = key;
    }
  }
}

namespace {

struct SortVariant {
  int waves, ipt;
  void (*hist)(const uint64_t*, int64_t, int, int32_t*, int64_t);
  void (*scat)(const uint64_t*, int64_t, int, const int32_t*, int64_t,
               uint64_t*);
};

template <int WAVES, int IPT>
constexpr SortVariant make_variant() {
  return {WAVES, IPT, rs_histogram_k<WAVES, IPT>, rs_scatter_k<WAVES, IPT>};
}

// Variants selectable via DE_SORT_VARIANT for measurement
// (tools/bench_sort3.py sweeps these vs rocPRIM).
const SortVariant kVariants[] = {
    make_variant<4, 4>(),   // 0: tile 1024
    make_variant<4, 16>(),  // 1: tile 4096 (round-1 geometry)
    make_variant<8, 8>(),   // 2: tile 4096, 512 threads
    make_variant<4, 8>(),   // 3: tile 2048
    make_variant<8, 4>(),   // 4: tile 2048, 512 threads
    make_variant<2, 4>(),   // 5: tile 512
    make_variant<8, 16>(),  // 6: tile 8192, 512 threads
    make_variant<16, 8>(),  // 7: tile 8192, 1024 threads
};

int env_variant() {
  static int v = [] {
    const char* e = std::getenv("DE_SORT_VARIANT");
    if (!e) return -1;
    int i = std::atoi(e);
    const int nv = (int)(sizeof(kVariants) / sizeof(kVariants[0]));
    return (i >= 0 && i < nv) ? i : -1;
  }();
  return v;
}

const SortVariant& pick_variant(int64_t n) {
  const int e = env_variant();
  if (e >= 0) return kVariants[e];
  // measured on MI355X (profiles/sort_sweep.md): small-tile 8-wave wins at
  // DLRM-26-table sizes (needs blocks >> 256 CUs; 178us vs rocPRIM 195us);
  // the 16-wave/8192 tile amortizes best once n is large (506us vs 502us
  // at 1.7M keys — parity)
  return n <= 512 * 1024 ? kVariants[4] : kVariants[7];
}

}  // namespace

void custom_radix_sort_keys(const uint64_t* keys_in, uint64_t* keys_out,
                            uint64_t* keys_tmp, int32_t* hist,
                            int32_t* scan_sums, int64_t n, int begin_bit,
                            int end_bit, hipStream_t stream) {
  const SortVariant& var = pick_variant(n);
  const int tile = var.waves * WAVE * var.ipt;
  const int64_t nblocks = rs_cdiv(n, tile);
  const int passes = (end_bit - begin_bit + 7) / 8;
  const int64_t m = (int64_t)RS_RADIX * nblocks;
  const int chunk = (int)rs_cdiv(m, 2048) < 64 ? 64 : (int)rs_cdiv(m, 2048);
  const int nb_scan = (int)rs_cdiv(m, chunk);

  const uint64_t* kin = keys_in;
  uint64_t* kout;
  bool to_tmp = (passes % 2) == 0;
  for (int p = 0; p < passes; ++p) {
    kout = to_tmp ? keys_tmp : keys_out;
    const int shift = begin_bit + p * 8;
    hipLaunchKernelGGL(var.hist, dim3((int)nblocks),
                       dim3(var.waves * WAVE), 0, stream, kin, n, shift, hist,
                       nblocks);
    hipLaunchKernelGGL(rs_scan_partials, dim3(nb_scan), dim3(WAVE), 0, stream,
                       hist, m, scan_sums, hist, chunk);
    hipLaunchKernelGGL(rs_scan_top, dim3(1), dim3(WAVE), 0, stream, scan_sums,
                       nb_scan);
    hipLaunchKernelGGL(rs_scan_addback, dim3(nb_scan), dim3(256), 0, stream,
                       hist, m, scan_sums, chunk);
    hipLaunchKernelGGL(var.scat, dim3((int)nblocks),
                       dim3(var.waves * WAVE), 0, stream, kin, n, shift, hist,
                       nblocks, kout);
    kin = kout;
    to_tmp = !to_tmp;
  }
}

size_t custom_radix_sort_hist_elems(int64_t n) {
  // sized for the smallest tile any variant uses (most blocks)
  int min_tile = kVariants[0].waves * WAVE * kVariants[0].ipt;
  for (const auto& v : kVariants) {
    const int t = v.waves * WAVE * v.ipt;
    if (t < min_tile) min_tile = t;
  }
  return (size_t)RS_RADIX * rs_cdiv(n, min_tile);
}
