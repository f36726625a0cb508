// Fused DLRM pairwise-dot interaction for gfx950 (MFMA bf16).
//
// forward:  feats [B, F, D] bf16 (feature 0 = bottom-MLP output)
//        -> out [B, OUT] bf16 = [tril(feats @ feats^T) | feats[:,0,:] | 0-pad]
// backward: gout [B, OUT] bf16 -> gfeats [B, F, D] bf16
//           (gfeats = (G + G^T) @ feats, + bottom-concat grad into row 0)
//
// Replaces the reference's stack + bmm + tril boolean-mask + concat chain
// (examples/dlrm/utils.py:92-113) with one kernel each way: per sample the
// 32x32 Gram tile is 12 v_mfma_f32_16x16x32_bf16 ops (lower-triangle tiles
// only), staged through an LDS copy of the sample's [32, D] feature block.
// Requirements: F <= 32, D % 32 == 0, bf16 inputs.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include "ops_api.h"

#define WAVE 64

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// A/B fragment gather for v_mfma_f32_16x16x32_bf16:
//   A[row = lane%16][k = (lane/16)*8 + r],  B[k][col = lane%16] — r in [0,8).
// C/D: col = lane&15, row = (lane>>4)*4 + reg.
// (cdna_hip_programming.md §3; verified on hardware by the asymmetric
// bmm-oracle test in tests/test_gpu.py.)

// Forward: one wave per sample.
template <int FMAX>  // padded feature count (32)
__global__ void dot_interact_fwd(const __hip_bfloat16* __restrict__ feats,
                                 __hip_bfloat16* __restrict__ out, int64_t B,
                                 int F, int D, int out_w, int tri_n) {
  extern __shared__ short lds_all[];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & (WAVE - 1);
  const int ldst = D + 8;  // +16B row pad: avoids 256B-row bank conflicts
  short* lds = lds_all + wave * FMAX * ldst;  // [FMAX][D+8] bf16 (as short)
  const int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) >> 6;
  const int64_t wave_id = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;

  for (int64_t b = wave_id; b < B; b += n_waves) {
    const short* src = reinterpret_cast<const short*>(feats) + b * (int64_t)F * D;
    // stage sample into LDS (pad rows >= F with zeros), 8 bf16 per lane-step
    const int total8 = FMAX * D / 8;
    for (int i = lane; i < total8; i += WAVE) {
      const int elem = i * 8;
      const int row = elem / D, col = elem % D;
      if (elem < F * D) {
        *reinterpret_cast<bf16x8*>(&lds[row * ldst + col]) =
            *reinterpret_cast<const bf16x8*>(&src[elem]);
      } else {
        bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        *reinterpret_cast<bf16x8*>(&lds[row * ldst + col]) = z;
      }
    }
    // per-wave LDS: hipcc inserts the lgkm waits for its own ds ops

    const int r16 = lane & 15;      // fragment row/col within 16
    const int khalf = lane >> 4;    // 0..3 -> k = khalf*8 + r

    // lower-triangle 16x16 tiles: (0,0), (1,0), (1,1)
    const int tiles_mi[3] = {0, 1, 1};
    const int tiles_ni[3] = {0, 0, 1};
#pragma unroll
    for (int t = 0; t < 3; ++t) {
      const int mi = tiles_mi[t], ni = tiles_ni[t];
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
      for (int k0 = 0; k0 < D; k0 += 32) {
        // a_frag: rows of tile mi; b_frag: rows of tile ni (B = A^T)
        bf16x8 a = *reinterpret_cast<const bf16x8*>(
            &lds[(mi * 16 + r16) * ldst + k0 + khalf * 8]);
        bf16x8 bfr = *reinterpret_cast<const bf16x8*>(
            &lds[(ni * 16 + r16) * ldst + k0 + khalf * 8]);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr, acc, 0, 0, 0);
      }
      // scatter lower-triangle entries
      __hip_bfloat16* orow = out + b * (int64_t)out_w;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int i = mi * 16 + (lane >> 4) * 4 + reg;
        const int j = ni * 16 + (lane & 15);
        if (i > j && i < F && j < F) {
          orow[i * (i - 1) / 2 + j] = __hip_bfloat16(acc[reg]);
        }
      }
    }
    // bottom-MLP re-concat + zero pad (bit copy of bf16 pattern)
    short* orow_s = reinterpret_cast<short*>(out + b * (int64_t)out_w);
    for (int c = lane; c < D; c += WAVE) {
      orow_s[tri_n + c] = lds[c];
    }
    for (int c = tri_n + D + lane; c < out_w; c += WAVE) {
      orow_s[c] = 0;
    }
  }
}

// Backward: one wave per sample.  gfeats = (G + G^T) @ feats + bottom grad.
template <int FMAX>
__global__ void dot_interact_bwd(const __hip_bfloat16* __restrict__ gout,
                                 const __hip_bfloat16* __restrict__ feats,
                                 __hip_bfloat16* __restrict__ gfeats, int64_t B,
                                 int F, int D, int out_w, int tri_n) {
  extern __shared__ short lds_all[];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & (WAVE - 1);
  // layout per wave: [FMAX][D] feats  +  [FMAX][FMAX] G_sym
  short* lds = lds_all + wave * (FMAX * D + FMAX * FMAX);
  short* gsym = lds + FMAX * D;
  const int64_t n_waves = ((int64_t)gridDim.x * blockDim.x) >> 6;
  const int64_t wave_id = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) >> 6;

  for (int64_t b = wave_id; b < B; b += n_waves) {
    const short* src = reinterpret_cast<const short*>(feats) + b * (int64_t)F * D;
    const int total8 = FMAX * D / 8;
    for (int i = lane; i < total8; i += WAVE) {
      const int elem = i * 8;
      if (elem < F * D) {
        *reinterpret_cast<bf16x8*>(&lds[elem]) =
            *reinterpret_cast<const bf16x8*>(&src[elem]);
      } else {
        bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        *reinterpret_cast<bf16x8*>(&lds[elem]) = z;
      }
    }
    // build G_sym [FMAX][FMAX] from the tril grad
    const __hip_bfloat16* grow = gout + b * (int64_t)out_w;
    for (int idx = lane; idx < FMAX * FMAX; idx += WAVE) {
      const int i = idx / FMAX, j = idx % FMAX;
      float g = 0.f;
      if (i < F && j < F && i != j) {
        const int r = i > j ? i : j, c = i > j ? j : i;
        g = float(grow[r * (r - 1) / 2 + c]);
      }
      __hip_bfloat16 hb(g);
      gsym[idx] = *reinterpret_cast<short*>(&hb);
    }

    const int r16 = lane & 15;
    const int khalf = lane >> 4;
    __hip_bfloat16* gf = gfeats + b * (int64_t)F * D;

    // grad_A = G_sym @ A : M=FMAX rows, N=D cols, K=FMAX (=32: one K step)
    for (int nj = 0; nj < D / 16; ++nj) {
#pragma unroll
      for (int mi = 0; mi < FMAX / 16; ++mi) {
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int k0 = 0; k0 < FMAX; k0 += 32) {
          // A-op = G_sym[mi*16 + r16][k0 + khalf*8 + r]
          bf16x8 a = *reinterpret_cast<const bf16x8*>(
              &gsym[(mi * 16 + r16) * FMAX + k0 + khalf * 8]);
          // B-op[k][col] = feats[k][nj*16 + r16]: k = k0 + khalf*8 + r
          bf16x8 bfr;
#pragma unroll
          for (int r = 0; r < 8; ++r) {
            bfr[r] = lds[(k0 + khalf * 8 + r) * D + nj * 16 + r16];
          }
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr, acc, 0, 0, 0);
        }
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int i = mi * 16 + (lane >> 4) * 4 + reg;
          const int j = nj * 16 + (lane & 15);
          if (i < F && j < D) {
            float v = acc[reg];
            if (i == 0) v += float(grow[tri_n + j]);  // bottom-concat grad
            gf[i * D + j] = __hip_bfloat16(v);
          }
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Packed variants: feats live as TWO tensors — bottom [B, D] (feature 0) and
// packed [P, B, D] (P = F-1 embedding outputs, feature-major).  The packed
// block is exactly the fused-group lookup output (world==1) or the mp->dp
// all-to-all recv buffer (world>1) VIEWED in place, so the torch.stack /
// per-pair split+merge copies disappear from the step.  `perm[f-1]` maps
// feature f (model input order) to its packed row (worker order) — a small
// device-resident buffer built once from the plan, keeping the interaction
// column order identical at every world size.
// ---------------------------------------------------------------------------

// sb/sp: element-row strides of the packed block — feature-major [P,B,D]
// uses (sb=1, sp=B); sample-major [B,P,D] uses (sb=P, sp=1).
//
// Block-cooperative IO: each workgroup (WPB waves) handles WPB consecutive
// samples.  In the feature-major layout the WPB samples' rows are ADJACENT
// within each feature plane, so a cooperative load moves WPB*256 B
// contiguous per plane instead of per-wave 256 B scatters (measured 2-3x
// kernel time on the scattered variant).  The backward additionally stages
// its [F, D] grad in LDS (overwriting nothing the MFMA still needs) and
// writes it back cooperatively the same way.

// waves (= samples) per block: templated; DI_WPB is the measured default,
// DE_DI_WPB env selects 2/4/8 for measurement (tools/bench_interact.py)
// measured per-direction optima (tools/bench_interact.py sweep): fwd 2,
// bwd 8 — the deltas are small; DE_DI_WPB overrides both for measurement
static int di_wpb(int dflt) {
  static int v = [] {
    const char* e = getenv("DE_DI_WPB");
    if (!e) return 0;
    int i = atoi(e);
    return (i == 2 || i == 4 || i == 8) ? i : 0;
  }();
  return v ? v : dflt;
}

template <int FMAX, int WPB>
__global__ void dot_interact_fwd_packed(
    const __hip_bfloat16* __restrict__ bottom,
    const __hip_bfloat16* __restrict__ packed,
    const int* __restrict__ perm, __hip_bfloat16* __restrict__ out, int64_t B,
    int F, int D, int out_w, int tri_n, int64_t sb, int64_t sp) {
  extern __shared__ short lds_all[];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & (WAVE - 1);
  const int tid = threadIdx.x;
  const int ldst = D + 8;
  short* lds = lds_all + wave * FMAX * ldst;
  const int d8 = D / 8;

  for (int64_t b0 = (int64_t)blockIdx.x * WPB; b0 < B;
       b0 += (int64_t)gridDim.x * WPB) {
    // cooperative load: chunk i -> (row, sample w, col); per row the WPB
    // samples' segments are contiguous in the feature-major layout
    for (int i = tid; i < FMAX * WPB * d8; i += WPB * WAVE) {
      const int row = i / (WPB * d8);
      const int rem = i % (WPB * d8);
      const int w = rem / d8;
      const int col = (rem % d8) * 8;
      const int64_t b = b0 + w;
      bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
      if (b < B) {
        if (row == 0) {
          v = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const short*>(bottom) + b * (int64_t)D + col);
        } else if (row < F) {
          v = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const short*>(packed) +
              ((int64_t)perm[row - 1] * sp + b * sb) * (int64_t)D + col);
        }
      }
      *reinterpret_cast<bf16x8*>(
          &lds_all[w * FMAX * ldst + row * ldst + col]) = v;
    }
    __syncthreads();

    const int64_t b = b0 + wave;
    if (b < B) {
      const int r16 = lane & 15;
      const int khalf = lane >> 4;
      const int tiles_mi[3] = {0, 1, 1};
      const int tiles_ni[3] = {0, 0, 1};
#pragma unroll
      for (int t = 0; t < 3; ++t) {
        const int mi = tiles_mi[t], ni = tiles_ni[t];
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        for (int k0 = 0; k0 < D; k0 += 32) {
          bf16x8 a = *reinterpret_cast<const bf16x8*>(
              &lds[(mi * 16 + r16) * ldst + k0 + khalf * 8]);
          bf16x8 bfr = *reinterpret_cast<const bf16x8*>(
              &lds[(ni * 16 + r16) * ldst + k0 + khalf * 8]);
          acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr, acc, 0, 0, 0);
        }
        __hip_bfloat16* orow = out + b * (int64_t)out_w;
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int i = mi * 16 + (lane >> 4) * 4 + reg;
          const int j = ni * 16 + (lane & 15);
          if (i > j && i < F && j < F) {
            orow[i * (i - 1) / 2 + j] = __hip_bfloat16(acc[reg]);
          }
        }
      }
      short* orow_s = reinterpret_cast<short*>(out + b * (int64_t)out_w);
      for (int c = lane; c < D; c += WAVE) {
        orow_s[tri_n + c] = lds[c];
      }
      for (int c = tri_n + D + lane; c < out_w; c += WAVE) {
        orow_s[c] = 0;
      }
    }
    __syncthreads();
  }
}

template <int FMAX, int WPB>
__global__ void dot_interact_bwd_packed(
    const __hip_bfloat16* __restrict__ gout,
    const __hip_bfloat16* __restrict__ bottom,
    const __hip_bfloat16* __restrict__ packed, const int* __restrict__ perm,
    __hip_bfloat16* __restrict__ gbottom, __hip_bfloat16* __restrict__ gpacked,
    int64_t B, int F, int D, int out_w, int tri_n, int64_t sb, int64_t sp) {
  extern __shared__ short lds_all[];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & (WAVE - 1);
  const int tid = threadIdx.x;
  // per wave: [FMAX][D] feats + [FMAX][FMAX] gsym.  The feats region is
  // REUSED for the grad: each nj column chunk is overwritten only after
  // both mi MFMA passes buffered their results in registers (keeps LDS at
  // 40KB/workgroup -> 4 workgroups/CU).
  const int per_wave = FMAX * D + FMAX * FMAX;
  short* lds = lds_all + wave * per_wave;
  short* gsym = lds + FMAX * D;
  const int d8 = D / 8;

  for (int64_t b0 = (int64_t)blockIdx.x * WPB; b0 < B;
       b0 += (int64_t)gridDim.x * WPB) {
    for (int i = tid; i < FMAX * WPB * d8; i += WPB * WAVE) {
      const int row = i / (WPB * d8);
      const int rem = i % (WPB * d8);
      const int w = rem / d8;
      const int col = (rem % d8) * 8;
      const int64_t b = b0 + w;
      bf16x8 v = {0, 0, 0, 0, 0, 0, 0, 0};
      if (b < B) {
        if (row == 0) {
          v = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const short*>(bottom) + b * (int64_t)D + col);
        } else if (row < F) {
          v = *reinterpret_cast<const bf16x8*>(
              reinterpret_cast<const short*>(packed) +
              ((int64_t)perm[row - 1] * sp + b * sb) * (int64_t)D + col);
        }
      }
      *reinterpret_cast<bf16x8*>(
          &lds_all[w * per_wave + row * D + col]) = v;
    }
    __syncthreads();

    const int64_t b = b0 + wave;
    if (b < B) {
      const __hip_bfloat16* grow = gout + b * (int64_t)out_w;
      for (int idx = lane; idx < FMAX * FMAX; idx += WAVE) {
        const int i = idx / FMAX, j = idx % FMAX;
        float g = 0.f;
        if (i < F && j < F && i != j) {
          const int r = i > j ? i : j, c = i > j ? j : i;
          g = float(grow[r * (r - 1) / 2 + c]);
        }
        __hip_bfloat16 hb(g);
        gsym[idx] = *reinterpret_cast<short*>(&hb);
      }

      const int r16 = lane & 15;
      const int khalf = lane >> 4;
      for (int nj = 0; nj < D / 16; ++nj) {
        f32x4 accs[FMAX / 16];
#pragma unroll
        for (int mi = 0; mi < FMAX / 16; ++mi) {
          f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
          for (int k0 = 0; k0 < FMAX; k0 += 32) {
            bf16x8 a = *reinterpret_cast<const bf16x8*>(
                &gsym[(mi * 16 + r16) * FMAX + k0 + khalf * 8]);
            bf16x8 bfr;
#pragma unroll
            for (int r = 0; r < 8; ++r) {
              bfr[r] = lds[(k0 + khalf * 8 + r) * D + nj * 16 + r16];
            }
            acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfr, acc, 0, 0, 0);
          }
          accs[mi] = acc;
        }
        // both mi passes read columns nj*16.. — now overwrite them in place
#pragma unroll
        for (int mi = 0; mi < FMAX / 16; ++mi) {
#pragma unroll
          for (int reg = 0; reg < 4; ++reg) {
            const int i = mi * 16 + (lane >> 4) * 4 + reg;
            const int j = nj * 16 + (lane & 15);
            float v = accs[mi][reg];
            if (i == 0) v += float(grow[tri_n + j]);
            __hip_bfloat16 hb(v);
            lds[i * D + j] = *reinterpret_cast<short*>(&hb);
          }
        }
      }
    }
    __syncthreads();

    // cooperative write-back: same (row, sample, col) mapping as the load
    for (int i = tid; i < FMAX * WPB * d8; i += WPB * WAVE) {
      const int row = i / (WPB * d8);
      const int rem = i % (WPB * d8);
      const int w = rem / d8;
      const int col = (rem % d8) * 8;
      const int64_t b = b0 + w;
      if (b >= B || row >= F) continue;
      const bf16x8 v = *reinterpret_cast<const bf16x8*>(
          &lds_all[w * per_wave + row * D + col]);
      if (row == 0) {
        *reinterpret_cast<bf16x8*>(
            reinterpret_cast<short*>(gbottom) + b * (int64_t)D + col) = v;
      } else {
        *reinterpret_cast<bf16x8*>(
            reinterpret_cast<short*>(gpacked) +
            ((int64_t)perm[row - 1] * sp + b * sb) * (int64_t)D + col) = v;
      }
    }
    __syncthreads();
  }
}

void launch_dot_interact_fwd_packed(const void* bottom, const void* packed,
                                    const int* perm, void* out, int64_t B,
                                    int F, int D, int out_w, int tri_n,
                                    int64_t sb, int64_t sp,
                                    hipStream_t stream) {
  const int wpb = di_wpb(2);
  const int block = wpb * WAVE;
  int64_t blocks = (B + wpb - 1) / wpb;
  if (blocks > 32768) blocks = 32768;
  const size_t lds = (size_t)wpb * 32 * (D + 8) * sizeof(short);
#define LFWD(W)                                                                \
  hipLaunchKernelGGL((dot_interact_fwd_packed<32, W>), dim3((int)blocks),      \
                     dim3(block), lds, stream,                                 \
                     (const __hip_bfloat16*)bottom,                            \
                     (const __hip_bfloat16*)packed, perm, (__hip_bfloat16*)out,\
                     B, F, D, out_w, tri_n, sb, sp)
  if (wpb == 2) LFWD(2); else if (wpb == 8) LFWD(8); else LFWD(4);
#undef LFWD
}

void launch_dot_interact_bwd_packed(const void* gout, const void* bottom,
                                    const void* packed, const int* perm,
                                    void* gbottom, void* gpacked, int64_t B,
                                    int F, int D, int out_w, int tri_n,
                                    int64_t sb, int64_t sp,
                                    hipStream_t stream) {
  const int wpb = di_wpb(8);
  const int block = wpb * WAVE;
  int64_t blocks = (B + wpb - 1) / wpb;
  if (blocks > 32768) blocks = 32768;
  const size_t lds = (size_t)wpb * (32 * D + 32 * 32) * sizeof(short);
#define LBWD(W)                                                                \
  hipLaunchKernelGGL((dot_interact_bwd_packed<32, W>), dim3((int)blocks),      \
                     dim3(block), lds, stream, (const __hip_bfloat16*)gout,    \
                     (const __hip_bfloat16*)bottom,                            \
                     (const __hip_bfloat16*)packed, perm,                      \
                     (__hip_bfloat16*)gbottom, (__hip_bfloat16*)gpacked, B, F, \
                     D, out_w, tri_n, sb, sp)
  if (wpb == 2) LBWD(2); else if (wpb == 8) LBWD(8); else LBWD(4);
#undef LBWD
}

void launch_dot_interact_fwd(const void* feats, void* out, int64_t B, int F,
                             int D, int out_w, int tri_n, hipStream_t stream) {
  const int block = 256;
  const int waves = block / WAVE;
  int64_t blocks = (B + waves - 1) / waves;
  if (blocks > 8192) blocks = 8192;
  const size_t lds = (size_t)waves * 32 * (D + 8) * sizeof(short);
  hipLaunchKernelGGL((dot_interact_fwd<32>), dim3((int)blocks), dim3(block),
                     lds, stream, (const __hip_bfloat16*)feats,
                     (__hip_bfloat16*)out, B, F, D, out_w, tri_n);
}

void launch_dot_interact_bwd(const void* gout, const void* feats, void* gfeats,
                             int64_t B, int F, int D, int out_w, int tri_n,
                             hipStream_t stream) {
  const int block = 256;
  const int waves = block / WAVE;
  int64_t blocks = (B + waves - 1) / waves;
  if (blocks > 8192) blocks = 8192;
  const size_t lds = (size_t)waves * (32 * D + 32 * 32) * sizeof(short);
  hipLaunchKernelGGL((dot_interact_bwd<32>), dim3((int)blocks), dim3(block),
                     lds, stream, (const __hip_bfloat16*)gout,
                     (const __hip_bfloat16*)feats, (__hip_bfloat16*)gfeats, B,
                     F, D, out_w, tri_n);
}
