// dfann — gfx950 (MI355X/CDNA4) kernels for the sharded ANN hot path.
//
// Built from scratch for CDNA4: wave64 everywhere, MFMA f32
// (v_mfma_f32_32x32x2_f32) for the coarse-quantizer / flat distance GEMM,
// inverted-list scans with PQ LUTs / SQ codecs staged in LDS over packed
// HBM code reads, and a threshold-filtered LDS candidate buffer with
// block-wide bitonic selection for every top-k stage.
//
// Numeric contract with the CPU oracle (oracle/core.py header): list-scan
// accumulations are sequential over the reduced axis with fp contraction
// off, so distances are bitwise equal to the oracle given shared trained
// artifacts (PQ ADC, SQ8, SQfp16 paths). The IVF-Flat scan uses a
// 16-lane-tree reduction (documented tolerance path).
//
// Tie-break everywhere: (distance, ascending id); minimize-keys internally
// (IP negated), faiss sign conventions restored at the output edge.

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <stdint.h>
#include <type_traits>

#define DFANN_BLOCK 256
#define SEL_CAP 1024            // candidate buffer entries (k <= 512)
#define PAD_POS 0xFFFFFFFFu
#define DFANN_FLT_MAX 3.402823466e+38f

// ---------------------------------------------------------------------------
// slab-arena addressing: code images live in fixed-size slabs of
// (1 << rlog) rows each (DESIGN.md §2 memory plan: the arena grows by
// whole slabs and is recycled slab-by-slab during CSR rebuilds, so the
// engine never holds a 2x contiguous copy). `pos` stays the global CSR
// row index; kernels translate through the device slab-pointer table
// (a few hundred entries, L1-resident).
// ---------------------------------------------------------------------------

__device__ __forceinline__ const uint8_t *slab_row(
    const uint8_t *const *__restrict__ slabs, int rlog, long long pos,
    int stride) {
  return slabs[pos >> rlog] +
         (size_t)(pos & ((1LL << rlog) - 1)) * (size_t)stride;
}

__device__ __forceinline__ uint8_t *slab_row_mut(
    uint8_t *const *__restrict__ slabs, int rlog, long long pos, int stride) {
  return slabs[pos >> rlog] +
         (size_t)(pos & ((1LL << rlog) - 1)) * (size_t)stride;
}

typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) unsigned uint32x4v;

// 16-B non-temporal (evict-first) load; uint4 is a class type, so go
// through the ext-vector form the builtin accepts
__device__ __forceinline__ uint4 nt_load16(const void *p) {
  uint32x4v v =
      __builtin_nontemporal_load(reinterpret_cast<const uint32x4v *>(p));
  union { uint32x4v v; uint4 u; } c;
  c.v = v;
  return c.u;
}

// ---------------------------------------------------------------------------
// selection machinery: threshold-filtered append + block bitonic compact
// LDS carve (16B aligned): float selD[SEL_CAP]; unsigned selP[SEL_CAP];
// int ctrl[4] = {cnt, pad, thrP, 0}; float thrD in ctrl-adjacent slot.
// ---------------------------------------------------------------------------

struct Sel {
  float *d;       // SEL_CAP
  unsigned *p;    // SEL_CAP
  int *cnt;
  float *thrD;
  unsigned *thrP;
};

__device__ __forceinline__ Sel sel_carve(char *base) {
  Sel s;
  s.d = reinterpret_cast<float *>(base);
  s.p = reinterpret_cast<unsigned *>(base + SEL_CAP * 4);
  s.cnt = reinterpret_cast<int *>(base + SEL_CAP * 8);
  s.thrD = reinterpret_cast<float *>(base + SEL_CAP * 8 + 4);
  s.thrP = reinterpret_cast<unsigned *>(base + SEL_CAP * 8 + 8);
  return s;
}
#define SEL_LDS_BYTES (SEL_CAP * 8 + 16)

__device__ __forceinline__ void sel_init(Sel s) {
  if (threadIdx.x == 0) {
    *s.cnt = 0;
    *s.thrD = DFANN_FLT_MAX;
    *s.thrP = PAD_POS;
  }
  for (int i = threadIdx.x; i < SEL_CAP; i += blockDim.x) {
    s.d[i] = DFANN_FLT_MAX;
    s.p[i] = PAD_POS;
  }
}

// lexicographic (dist, pos) — pads (FLT_MAX, PAD_POS) sort last
__device__ __forceinline__ bool sel_less(float d0, unsigned p0, float d1, unsigned p1) {
  return d0 < d1 || (d0 == d1 && p0 < p1);
}

__device__ __forceinline__ void sel_try(Sel s, float dist, unsigned pos) {
  // caller guarantees capacity headroom; threshold is stable between compacts
  if (sel_less(dist, pos, *s.thrD, *s.thrP)) {
    int slot = atomicAdd(s.cnt, 1);
    s.d[slot] = dist;
    s.p[slot] = pos;
  }
}

// block-wide: pad [cnt, CAP), bitonic sort CAP entries, truncate to k.
// Requires a preceding __syncthreads() by the caller.
__device__ void sel_compact(Sel s, int k) {
  int cnt = *s.cnt;
  __syncthreads();
  for (int i = threadIdx.x; i < SEL_CAP; i += blockDim.x) {
    if (i >= cnt) {
      s.d[i] = DFANN_FLT_MAX;
      s.p[i] = PAD_POS;
    }
  }
  for (int kk = 2; kk <= SEL_CAP; kk <<= 1) {
    for (int j = kk >> 1; j > 0; j >>= 1) {
      __syncthreads();
      for (int i = threadIdx.x; i < SEL_CAP; i += blockDim.x) {
        int ixj = i ^ j;
        if (ixj > i) {
          bool asc = ((i & kk) == 0);
          float di = s.d[i], dj = s.d[ixj];
          unsigned pi = s.p[i], pj = s.p[ixj];
          bool swap_needed = asc ? sel_less(dj, pj, di, pi) : sel_less(di, pi, dj, pj);
          if (swap_needed) {
            s.d[i] = dj; s.d[ixj] = di;
            s.p[i] = pj; s.p[ixj] = pi;
          }
        }
      }
    }
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    int newcnt = cnt < k ? cnt : k;
    *s.cnt = newcnt;
    if (newcnt >= k) {
      *s.thrD = s.d[k - 1];
      *s.thrP = s.p[k - 1];
    }
  }
  __syncthreads();
}

// returns true if a compact happened (callers re-sync on their own)
__device__ __forceinline__ void sel_guard(Sel s, int k, int max_appends) {
  __syncthreads();
  if (*s.cnt > SEL_CAP - max_appends) sel_compact(s, k);
}

// ---------------------------------------------------------------------------
// register-resident per-thread top-K (K <= 16) — the fast selection path.
// Each thread keeps an ascending (dist, pos) array in registers (static
// indices via full unroll); most elements fail the single arr[K-1] compare.
// Block-wide result: K extraction rounds over the threads' heads.
// ---------------------------------------------------------------------------

template <int K>
struct RegTopK {
  float d[K];
  unsigned p[K];

  __device__ __forceinline__ void init() {
#pragma unroll
    for (int i = 0; i < K; ++i) {
      d[i] = DFANN_FLT_MAX;
      p[i] = PAD_POS;
    }
  }

  __device__ __forceinline__ void push(float nd, unsigned np) {
    if (!sel_less(nd, np, d[K - 1], p[K - 1])) return;
    d[K - 1] = nd;
    p[K - 1] = np;
#pragma unroll
    for (int i = K - 1; i > 0; --i) {
      if (sel_less(d[i], p[i], d[i - 1], p[i - 1])) {
        float td = d[i]; d[i] = d[i - 1]; d[i - 1] = td;
        unsigned tp = p[i]; p[i] = p[i - 1]; p[i - 1] = tp;
      }
    }
  }

  // drop the current head (after it won an extraction round)
  __device__ __forceinline__ void pop_head() {
#pragma unroll
    for (int i = 0; i < K - 1; ++i) {
      d[i] = d[i + 1];
      p[i] = p[i + 1];
    }
    d[K - 1] = DFANN_FLT_MAX;
    p[K - 1] = PAD_POS;
  }
};

// Block-wide merge of per-thread RegTopK heads. Phase 1: each wave
// extracts its own top-k with pure shfl rounds (no barriers); phase 2:
// one barrier, then thread 0 merges the per-wave sorted lists.
// lds: 8*16*8 = 1 KiB scratch. Supports blockDim 256 or 512 (4/8 waves).
#define REGSEL_LDS_BYTES 1152
template <int K>
__device__ void regtopk_block_extract(RegTopK<K> &loc, int k, char *lds,
                                      float *out_d, unsigned *out_p) {
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  int nw = blockDim.x >> 6;  // 4 or 8 waves
  float *wvd = reinterpret_cast<float *>(lds);             // [8][K]
  unsigned *wvp = reinterpret_cast<unsigned *>(lds + 8 * K * 4);
  for (int round = 0; round < k; ++round) {
    float cd = loc.d[0];
    unsigned cp = loc.p[0];
    unsigned cl = (unsigned)lane;
#pragma unroll
    for (int o = 32; o > 0; o >>= 1) {
      float od = __shfl_down(cd, o, 64);
      unsigned op = __shfl_down(cp, o, 64);
      unsigned ol = __shfl_down(cl, o, 64);
      if (sel_less(od, op, cd, cp)) { cd = od; cp = op; cl = ol; }
    }
    cd = __shfl(cd, 0, 64);
    cp = __shfl(cp, 0, 64);
    cl = __shfl(cl, 0, 64);
    if (lane == 0) {
      wvd[w * K + round] = cd;
      wvp[w * K + round] = cp;
    }
    if ((unsigned)lane == cl) loc.pop_head();
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    int cur[8] = {};
    for (int round = 0; round < k; ++round) {
      int bw = 0;
      float bd = DFANN_FLT_MAX;
      unsigned bp = PAD_POS;
#pragma unroll
      for (int v = 0; v < 8; ++v) {
        if (v < nw) {
          // a wave list holds exactly k entries; exhausted reads as pad
          float bv = cur[v] < k ? wvd[v * K + cur[v]] : DFANN_FLT_MAX;
          unsigned pv = cur[v] < k ? wvp[v * K + cur[v]] : PAD_POS;
          if (v == 0 || sel_less(bv, pv, bd, bp)) { bd = bv; bp = pv; bw = v; }
        }
      }
      out_d[round] = bd;
      out_p[round] = bp;
#pragma unroll
      for (int v = 0; v < 8; ++v)
        if (v == bw) ++cur[v];  // static index (rule 20: no scratch)
    }
  }
  __syncthreads();
}

// ---------------------------------------------------------------------------
// GEMM (fp32 MFMA): C[i][j] = sum_k A[i][k] * B[j][k]   (A: MxK, B: NxK)
// 128x128 tile, 4 waves x (2x2 of 32x32) on v_mfma_f32_32x32x2_f32.
// Operand map (cdna_hip_programming.md §3): lane l feeds A[i=l&31][k=l>>5],
// B[k=l>>5][j=l&31]; C/D: col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5).
// ---------------------------------------------------------------------------

#define GT 128
#define GK 32

__device__ __forceinline__ float gemm_key(float v, int row, int col,
                                          const float *qn, const float *bn,
                                          int mode) {
  if (mode == 0) return -v;
  if (mode == 1) return bn[col] - 2.0f * v;
  if (mode == 2) return (qn[row] - 2.0f * v) + bn[col];
  return v;
}

// mode: -1 raw ip; 0 key=-ip; 1 key=bn[col]-2ip; 2 key=(qn[row]-2ip)+bn[col]
// (fused epilogue)
extern "C" __global__ __launch_bounds__(256) void k_gemm_nt(
    const float *__restrict__ A, const float *__restrict__ B,
    float *__restrict__ C, int M, int N, int K, int lda, int ldb, int ldc,
    const float *__restrict__ qn, const float *__restrict__ bn, int mode) {
  __shared__ float sA[GT][GK + 1];
  __shared__ float sB[GT][GK + 1];
  int bi = blockIdx.y * GT;
  int bj = blockIdx.x * GT;
  int tid = threadIdx.x;
  int lane = tid & 63, w = tid >> 6;
  int wr = (w >> 1) * 64, wc = (w & 1) * 64;
  f32x16 acc00 = {}, acc01 = {}, acc10 = {}, acc11 = {};
  int rl = lane & 31;
  int klane = lane >> 5;
  for (int k0 = 0; k0 < K; k0 += GK) {
    for (int e = tid; e < GT * GK; e += 256) {
      int r = e >> 5, c = e & 31;  // GK == 32
      sA[r][c] = (bi + r < M && k0 + c < K) ? A[(size_t)(bi + r) * lda + k0 + c] : 0.f;
      sB[r][c] = (bj + r < N && k0 + c < K) ? B[(size_t)(bj + r) * ldb + k0 + c] : 0.f;
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < GK; kk += 2) {
      float a0 = sA[wr + rl][kk + klane];
      float a1 = sA[wr + 32 + rl][kk + klane];
      float b0 = sB[wc + rl][kk + klane];
      float b1 = sB[wc + 32 + rl][kk + klane];
      acc00 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b0, acc00, 0, 0, 0);
      acc01 = __builtin_amdgcn_mfma_f32_32x32x2f32(a0, b1, acc01, 0, 0, 0);
      acc10 = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b0, acc10, 0, 0, 0);
      acc11 = __builtin_amdgcn_mfma_f32_32x32x2f32(a1, b1, acc11, 0, 0, 0);
    }
    __syncthreads();
  }
  int row_in_tile_base = (lane >> 5) * 4;
  int col_in_tile = lane & 31;
#pragma unroll
  for (int rg = 0; rg < 16; ++rg) {
    int rit = (rg & 3) + 8 * (rg >> 2) + row_in_tile_base;
    {
      int row = bi + wr + rit, col = bj + wc + col_in_tile;
      if (row < M && col < N)
        C[(size_t)row * ldc + col] = gemm_key(acc00[rg], row, col, qn, bn, mode);
      col = bj + wc + 32 + col_in_tile;
      if (row < M && col < N)
        C[(size_t)row * ldc + col] = gemm_key(acc01[rg], row, col, qn, bn, mode);
      row = bi + wr + 32 + rit;
      col = bj + wc + col_in_tile;
      if (row < M && col < N)
        C[(size_t)row * ldc + col] = gemm_key(acc10[rg], row, col, qn, bn, mode);
      col = bj + wc + 32 + col_in_tile;
      if (row < M && col < N)
        C[(size_t)row * ldc + col] = gemm_key(acc11[rg], row, col, qn, bn, mode);
    }
  }
}

// ---------------------------------------------------------------------------
// bf16 GEMM: C[i][j] = sum_k A[i][k]*B[j][k], A/B bf16 (pre-converted),
// C fp32. v_mfma_f32_16x16x32_bf16, 128x128 tile, 4 waves x (4x4 of
// 16x16). Used for the APPROXIMATE assign/coarse path on huge nlist
// (spec "coarse_bf16") — ~16x the f32 MFMA rate.
// Fragment maps (verified on hardware by tests/test_gpu_parity.py):
//   A: lane l supplies A[i = l&15][k = (l>>4)*8 + e], e in 0..7
//   B: lane l supplies B[k = (l>>4)*8 + e][j = l&15]
//   C/D: col = lane&15, row = (lane>>4)*4 + reg, reg in 0..3
// ---------------------------------------------------------------------------

typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short bf16x8;

extern "C" __global__ void k_f32_to_bf16(const float *__restrict__ in,
                                         long long n,
                                         unsigned short *__restrict__ out) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; i < n; i += (long long)gridDim.x * blockDim.x) {
    union { float f; unsigned u; } v;
    v.f = in[i];
    // round-to-nearest-even bf16 truncation
    unsigned lsb = (v.u >> 16) & 1u;
    out[i] = (unsigned short)((v.u + 0x7FFFu + lsb) >> 16);
  }
}

#define GB_T 128
#define GB_K 64

// Double-buffered, T14-scheduled (cdna_hip_programming.md §5.5 T14 /
// Guideline 15): tile t+1's global loads are ISSUED before tile t's MFMA
// phase and their LDS write lands after, behind the other buffer — HBM
// latency hides under the MFMAs, one barrier per K-tile.
extern "C" __global__ __launch_bounds__(256) void k_gemm_bf16_nt(
    const unsigned short *__restrict__ A, const unsigned short *__restrict__ B,
    float *__restrict__ C, int M, int N, int K, int lda, int ldb, int ldc,
    const float *__restrict__ qn, const float *__restrict__ bn, int mode) {
  __shared__ unsigned short sA[2][GB_T][GB_K + 8];
  __shared__ unsigned short sB[2][GB_T][GB_K + 8];
  int bi = blockIdx.y * GB_T;
  int bj = blockIdx.x * GB_T;
  int tid = threadIdx.x;
  int lane = tid & 63, w = tid >> 6;
  int wr = (w >> 1) * 64, wc = (w & 1) * 64;
  f32x4 acc[4][4] = {};
  int li = lane & 15;
  int ke = (lane >> 4) * 8;
  // each thread stages 4 16-B groups per operand per tile
  int r_[4], c8_[4];
#pragma unroll
  for (int g = 0; g < 4; ++g) {
    int e8 = tid + g * 256;
    r_[g] = e8 >> 3;
    c8_[g] = (e8 & 7) * 8;
  }
  uint4 va[4], vb[4];

#define GB_LOAD(K0)                                                            \
  _Pragma("unroll") for (int g = 0; g < 4; ++g) {                              \
    int r = r_[g], c8 = c8_[g];                                                \
    uint4 x = {0, 0, 0, 0}, y = {0, 0, 0, 0};                                  \
    if (bi + r < M) {                                                          \
      if ((K0) + c8 + 7 < K) {                                                 \
        x = *reinterpret_cast<const uint4 *>(                                  \
            &A[(size_t)(bi + r) * lda + (K0) + c8]);                           \
      } else {                                                                 \
        unsigned short tmp[8] = {};                                            \
        for (int t = 0; t < 8; ++t)                                            \
          if ((K0) + c8 + t < K)                                               \
            tmp[t] = A[(size_t)(bi + r) * lda + (K0) + c8 + t];                \
        x = *reinterpret_cast<const uint4 *>(tmp);                             \
      }                                                                        \
    }                                                                          \
    if (bj + r < N) {                                                          \
      if ((K0) + c8 + 7 < K) {                                                 \
        y = *reinterpret_cast<const uint4 *>(                                  \
            &B[(size_t)(bj + r) * ldb + (K0) + c8]);                           \
      } else {                                                                 \
        unsigned short tmp[8] = {};                                            \
        for (int t = 0; t < 8; ++t)                                            \
          if ((K0) + c8 + t < K)                                               \
            tmp[t] = B[(size_t)(bj + r) * ldb + (K0) + c8 + t];                \
        y = *reinterpret_cast<const uint4 *>(tmp);                             \
      }                                                                        \
    }                                                                          \
    va[g] = x;                                                                 \
    vb[g] = y;                                                                 \
  }

#define GB_WRITE(BUF)                                                          \
  _Pragma("unroll") for (int g = 0; g < 4; ++g) {                              \
    *reinterpret_cast<uint4 *>(&sA[BUF][r_[g]][c8_[g]]) = va[g];               \
    *reinterpret_cast<uint4 *>(&sB[BUF][r_[g]][c8_[g]]) = vb[g];               \
  }

#define GB_MFMA(BUF)                                                           \
  _Pragma("unroll") for (int kk = 0; kk < GB_K; kk += 32) {                    \
    _Pragma("unroll") for (int ti = 0; ti < 4; ++ti) {                         \
      bf16x8 a0 = *reinterpret_cast<const bf16x8 *>(                           \
          &sA[BUF][wr + ti * 16 + li][kk + ke]);                               \
      _Pragma("unroll") for (int tj = 0; tj < 4; ++tj) {                       \
        bf16x8 b0 = *reinterpret_cast<const bf16x8 *>(                         \
            &sB[BUF][wc + tj * 16 + li][kk + ke]);                             \
        acc[ti][tj] =                                                          \
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[ti][tj],       \
                                                    0, 0, 0);                  \
      }                                                                        \
    }                                                                          \
  }

  GB_LOAD(0)
  GB_WRITE(0)
  __syncthreads();
  int cur = 0;
  for (int k0 = GB_K; k0 < K; k0 += GB_K) {
    GB_LOAD(k0)      // in flight during the MFMAs below
    GB_MFMA(cur)
    GB_WRITE(cur ^ 1)  // waits the loads; other buffer, so no barrier first
    __syncthreads();
    cur ^= 1;
  }
  GB_MFMA(cur)

  int rrow = (lane >> 4) * 4;
#pragma unroll
  for (int ti = 0; ti < 4; ++ti) {
#pragma unroll
    for (int tj = 0; tj < 4; ++tj) {
#pragma unroll
      for (int rg = 0; rg < 4; ++rg) {
        int row = bi + wr + ti * 16 + rrow + rg;
        int col = bj + wc + tj * 16 + li;
        if (row < M && col < N)
          C[(size_t)row * ldc + col] =
              gemm_key(acc[ti][tj][rg], row, col, qn, bn, mode);
      }
    }
  }
#undef GB_LOAD
#undef GB_WRITE
#undef GB_MFMA
}

// glds-staged variant (cdna_hip_programming.md §5 ladder step 3 /
// Guideline 15): global->LDS DMA (16-B wide) stages the next tile while
// the MFMAs run; ONE __shared__ object (a second one de-pipelines glds —
// §5 ".s-level traps" (a)); LDS image is lane-linear, so the bank
// swizzle lives on the SOURCE address and the fragment-read offset
// (rule 21): byte ^= (row&7)<<4 cuts the 16-lane ds_read_b128 group from
// 8-way to 2-way. Requires K % 8 == 0 (host falls back otherwise).
extern "C" __global__ __launch_bounds__(256) void k_gemm_bf16_glds(
    const unsigned short *__restrict__ A, const unsigned short *__restrict__ B,
    float *__restrict__ C, int M, int N, int K, int lda, int ldb, int ldc,
    const float *__restrict__ qn, const float *__restrict__ bn, int mode) {
  // [buf][operand][row][col] bf16, linear (swizzle on source/read)
  __shared__ unsigned short smem2[2][2][GB_T][GB_K];
  int bi = blockIdx.y * GB_T;
  int bj = blockIdx.x * GB_T;
  int tid = threadIdx.x;
  int lane = tid & 63, w = tid >> 6;
  int wr = (w >> 1) * 64, wc = (w & 1) * 64;
  f32x4 acc[4][4] = {};
  int li = lane & 15;
  int ke = (lane >> 4) * 8;

  // wave w stages rows [w*32, w*32+32) of each operand: 4 glds per operand,
  // each covering 64 consecutive 16-B groups (8 rows)
#define GLDS_STAGE(BUF, K0)                                                    \
  _Pragma("unroll") for (int i = 0; i < 4; ++i) {                              \
    int g = (w * 4 + i) * 64 + lane; /* group index, lane-linear */            \
    int r = g >> 3;                                                            \
    int c8 = (g & 7) * 8;                                                      \
    int c8s = c8 ^ ((r & 7) << 3); /* source-side XOR swizzle (elements) */    \
    unsigned short *ldst =                                                     \
        &smem2[BUF][0][0][0] + ((size_t)(w * 4 + i) * 64) * 8;                 \
    const unsigned short *ga = &A[(size_t)(bi + r) * lda + (K0) + c8s];        \
    const unsigned short *gb = &B[(size_t)(bj + r) * ldb + (K0) + c8s];        \
    bool oka = (bi + r < M) && ((K0) + c8s + 7 < K);                           \
    bool okb = (bj + r < N) && ((K0) + c8s + 7 < K);                           \
    if (oka)                                                                   \
      __builtin_amdgcn_global_load_lds(                                        \
          (const __attribute__((address_space(1))) unsigned int *)ga,          \
          (__attribute__((address_space(3))) unsigned int *)ldst, 16, 0, 0);   \
    else                                                                       \
      *reinterpret_cast<uint4 *>(                                              \
          &smem2[BUF][0][0][0] + (size_t)g * 8) = uint4{0, 0, 0, 0};           \
    unsigned short *ldstB =                                                    \
        &smem2[BUF][1][0][0] + ((size_t)(w * 4 + i) * 64) * 8;                 \
    if (okb)                                                                   \
      __builtin_amdgcn_global_load_lds(                                        \
          (const __attribute__((address_space(1))) unsigned int *)gb,          \
          (__attribute__((address_space(3))) unsigned int *)ldstB, 16, 0, 0);  \
    else                                                                       \
      *reinterpret_cast<uint4 *>(                                              \
          &smem2[BUF][1][0][0] + (size_t)g * 8) = uint4{0, 0, 0, 0};           \
  }

#define GLDS_MFMA(BUF)                                                         \
  _Pragma("unroll") for (int kk = 0; kk < GB_K; kk += 32) {                    \
    _Pragma("unroll") for (int ti = 0; ti < 4; ++ti) {                         \
      int ra = wr + ti * 16 + li;                                              \
      int ca = (kk + ke) ^ ((ra & 7) << 3);                                    \
      bf16x8 a0 = *reinterpret_cast<const bf16x8 *>(&smem2[BUF][0][ra][ca]);   \
      _Pragma("unroll") for (int tj = 0; tj < 4; ++tj) {                       \
        int rb = wc + tj * 16 + li;                                            \
        int cbx = (kk + ke) ^ ((rb & 7) << 3);                                 \
        bf16x8 b0 =                                                            \
            *reinterpret_cast<const bf16x8 *>(&smem2[BUF][1][rb][cbx]);        \
        acc[ti][tj] =                                                          \
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[ti][tj],       \
                                                    0, 0, 0);                  \
      }                                                                        \
    }                                                                          \
  }

  GLDS_STAGE(0, 0)
  __syncthreads();
  int cur = 0;
  for (int k0 = GB_K; k0 < K; k0 += GB_K) {
    GLDS_STAGE(cur ^ 1, k0)  // DMA in flight during the MFMAs
    GLDS_MFMA(cur)
    __syncthreads();  // drains the glds queue (vmcnt(0) inside)
    cur ^= 1;
  }
  GLDS_MFMA(cur)

  int rrow = (lane >> 4) * 4;
#pragma unroll
  for (int ti = 0; ti < 4; ++ti) {
#pragma unroll
    for (int tj = 0; tj < 4; ++tj) {
#pragma unroll
      for (int rg = 0; rg < 4; ++rg) {
        int row = bi + wr + ti * 16 + rrow + rg;
        int col = bj + wc + tj * 16 + li;
        if (row < M && col < N)
          C[(size_t)row * ldc + col] =
              gemm_key(acc[ti][tj][rg], row, col, qn, bn, mode);
      }
    }
  }
#undef GLDS_STAGE
#undef GLDS_MFMA
}

// ---------------------------------------------------------------------------
// row norms ||x_i||^2, one block per row
// ---------------------------------------------------------------------------

extern "C" __global__ __launch_bounds__(256) void k_rownorm(
    const float *__restrict__ x, long long n, int d, float *__restrict__ out) {
  long long row = blockIdx.x;
  if (row >= n) return;
  const float *xp = x + row * d;
  float acc = 0.f;
  for (int t = threadIdx.x; t < d; t += blockDim.x) {
    float v = xp[t];
    acc += v * v;
  }
#pragma unroll
  for (int o = 32; o > 0; o >>= 1) acc += __shfl_down(acc, o, 64);
  __shared__ float ws[4];
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  if (lane == 0) ws[w] = acc;
  __syncthreads();
  if (threadIdx.x == 0) out[row] = ws[0] + ws[1] + ws[2] + ws[3];
}

// ---------------------------------------------------------------------------
// top-k per row over a key matrix (stream-select), pos = base + col.
// out_d/out_p: (rows, k). grid.x = rows. dynamic LDS: SEL_LDS_BYTES.
// ---------------------------------------------------------------------------

extern "C" __global__ __launch_bounds__(256) void k_topk_rows(
    const float *__restrict__ keys, long long rows, long long cols,
    long long ldk, int k, unsigned base, long long ldo,
    float *__restrict__ out_d, unsigned *__restrict__ out_p) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  Sel s = sel_carve(smem);
  long long row = blockIdx.x;
  if (row >= rows) return;
  sel_init(s);
  __syncthreads();
  const float *kp = keys + row * ldk;
  const int BS = blockDim.x;
  if ((ldk & 3) == 0) {
    // float4 row reads (see k_topk_rows_rk) — Sel is order-invariant.
    // Appends are split into TWO guarded half-batches of 2*BS: a single
    // 4*BS batch can exceed the Sel headroom (SEL_CAP - k) at large k —
    // at k=512 the buffer overflowed its LDS carve (caught by
    // test_engine_caps_and_filtered_overfetch).
    const float4 *kp4 = reinterpret_cast<const float4 *>(kp);
    long long c4n = cols >> 2;
    for (long long c0 = 0; c0 < c4n; c0 += BS) {
      long long c4 = c0 + threadIdx.x;
      bool v = c4 < c4n;
      float4 v4 = v ? kp4[c4] : float4{0.f, 0.f, 0.f, 0.f};
      sel_guard(s, k, 2 * BS);
      if (v) {
        sel_try(s, v4.x, (unsigned)(c4 * 4 + 0) + base);
        sel_try(s, v4.y, (unsigned)(c4 * 4 + 1) + base);
      }
      sel_guard(s, k, 2 * BS);
      if (v) {
        sel_try(s, v4.z, (unsigned)(c4 * 4 + 2) + base);
        sel_try(s, v4.w, (unsigned)(c4 * 4 + 3) + base);
      }
    }
    sel_guard(s, k, BS);
    for (long long c = (c4n << 2) + threadIdx.x; c < cols; c += BS)
      sel_try(s, kp[c], (unsigned)c + base);
  } else {
    for (long long c0 = 0; c0 < cols; c0 += (long long)2 * BS) {
      sel_guard(s, k, 2 * BS);
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        long long c = c0 + u * BS + threadIdx.x;
        if (c < cols) sel_try(s, kp[c], (unsigned)c + base);
      }
    }
  }
  __syncthreads();
  sel_compact(s, k);
  int cnt = *s.cnt;
  for (int j = threadIdx.x; j < k; j += blockDim.x) {
    bool v = j < cnt;
    out_d[row * ldo + j] = v ? s.d[j] : DFANN_FLT_MAX;
    out_p[row * ldo + j] = v ? s.p[j] : PAD_POS;
  }
}

// register-path top-k per row (k <= 16): no LDS buffer, no bitonic
extern "C" __global__ __launch_bounds__(256) void k_topk_rows_rk(
    const float *__restrict__ keys, long long rows, long long cols,
    long long ldk, int k, unsigned base, long long ldo,
    float *__restrict__ out_d, unsigned *__restrict__ out_p) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  long long row = blockIdx.x;
  if (row >= rows) return;
  RegTopK<16> loc;
  loc.init();
  const float *kp = keys + row * ldk;
  // float4 row reads (1 KB per wave-load instead of 256 B) + 2 vectors
  // in flight: the 4 B/lane version capped the 8192x65536 coarse top-k
  // at ~1.5 TB/s. Lane->element assignment changes, result does not
  // (RegTopK extraction orders globally by (dist, id)).
  const long long BS = blockDim.x;
  if ((ldk & 3) == 0) {  // float4-aligned rows
    const float4 *kp4 = reinterpret_cast<const float4 *>(kp);
    long long c4n = cols >> 2;
    long long c4 = threadIdx.x;
    for (; c4 + BS < c4n; c4 += 2 * BS) {
      float4 v0 = kp4[c4], v1 = kp4[c4 + BS];
      loc.push(v0.x, (unsigned)(c4 * 4 + 0) + base);
      loc.push(v0.y, (unsigned)(c4 * 4 + 1) + base);
      loc.push(v0.z, (unsigned)(c4 * 4 + 2) + base);
      loc.push(v0.w, (unsigned)(c4 * 4 + 3) + base);
      loc.push(v1.x, (unsigned)((c4 + BS) * 4 + 0) + base);
      loc.push(v1.y, (unsigned)((c4 + BS) * 4 + 1) + base);
      loc.push(v1.z, (unsigned)((c4 + BS) * 4 + 2) + base);
      loc.push(v1.w, (unsigned)((c4 + BS) * 4 + 3) + base);
    }
    for (; c4 < c4n; c4 += BS) {
      float4 v0 = kp4[c4];
      loc.push(v0.x, (unsigned)(c4 * 4 + 0) + base);
      loc.push(v0.y, (unsigned)(c4 * 4 + 1) + base);
      loc.push(v0.z, (unsigned)(c4 * 4 + 2) + base);
      loc.push(v0.w, (unsigned)(c4 * 4 + 3) + base);
    }
    for (long long c = (c4n << 2) + threadIdx.x; c < cols; c += BS)
      loc.push(kp[c], (unsigned)c + base);
  } else {
    for (long long c = threadIdx.x; c < cols; c += BS)
      loc.push(kp[c], (unsigned)c + base);
  }
  __syncthreads();
  regtopk_block_extract<16>(loc, k, smem, out_d + row * ldo,
                            out_p + row * ldo);
}

// ---------------------------------------------------------------------------
// running argmin across key-matrix chunks (assignment). rows = points.
// best_v/best_i persist across chunk calls (init by k_fill_assign_init).
// Ties: lowest global column (chunks ascending, strict <).
// 256^2-tile variant (cdna_hip_programming.md §5 glds table: 256² BK=64
// 2-buffer glds is the top tier for large GEMMs — the 128² tile peaks
// ~620 TF, this shape ~1.2 PF): 512 threads = 8 waves as 2(M)x4(N),
// wave tile 128x64, acc 8x4 f32x4. Same source-XOR swizzle + lane-linear
// LDS + single __shared__ as k_gemm_bf16_glds. Used for the big
// coarse/assign GEMMs (N = nlist >= 2048); ragged M/N/K handled by the
// stage guards (out-of-range positions stage zeros).
#define GB2_T 256
extern "C" __global__ __launch_bounds__(512) void k_gemm_bf16_256(
    const unsigned short *__restrict__ A, const unsigned short *__restrict__ B,
    float *__restrict__ C, int M, int N, int K, int lda, int ldb, int ldc,
    const float *__restrict__ qn, const float *__restrict__ bn, int mode) {
  __shared__ unsigned short smem2[2][2][GB2_T][GB_K];
  int bi = blockIdx.y * GB2_T;
  int bj = blockIdx.x * GB2_T;
  int tid = threadIdx.x;
  int lane = tid & 63, w = tid >> 6;
  int wr = (w >> 2) * 128, wc = (w & 3) * 64;
  f32x4 acc[8][4] = {};
  int li = lane & 15;
  int ke = (lane >> 4) * 8;

  // 2048 16-B groups per operand, 8 waves x 4 glds x 64 lanes
#define GLDS2_STAGE(BUF, K0)                                                   \
  _Pragma("unroll") for (int i = 0; i < 4; ++i) {                              \
    int g = (w * 4 + i) * 64 + lane;                                           \
    int r = g >> 3;                                                            \
    int c8 = (g & 7) * 8;                                                      \
    int c8s = c8 ^ ((r & 7) << 3);                                             \
    unsigned short *ldst = &smem2[BUF][0][0][0] + ((size_t)(w * 4 + i) * 64) * 8; \
    const unsigned short *ga = &A[(size_t)(bi + r) * lda + (K0) + c8s];        \
    const unsigned short *gb = &B[(size_t)(bj + r) * ldb + (K0) + c8s];        \
    bool oka = (bi + r < M) && ((K0) + c8s + 7 < K);                           \
    bool okb = (bj + r < N) && ((K0) + c8s + 7 < K);                           \
    if (oka)                                                                   \
      __builtin_amdgcn_global_load_lds(                                        \
          (const __attribute__((address_space(1))) unsigned int *)ga,          \
          (__attribute__((address_space(3))) unsigned int *)ldst, 16, 0, 0);   \
    else                                                                       \
      *reinterpret_cast<uint4 *>(                                              \
          &smem2[BUF][0][0][0] + (size_t)g * 8) = uint4{0, 0, 0, 0};           \
    unsigned short *ldstB = &smem2[BUF][1][0][0] + ((size_t)(w * 4 + i) * 64) * 8; \
    if (okb)                                                                   \
      __builtin_amdgcn_global_load_lds(                                        \
          (const __attribute__((address_space(1))) unsigned int *)gb,          \
          (__attribute__((address_space(3))) unsigned int *)ldstB, 16, 0, 0);  \
    else                                                                       \
      *reinterpret_cast<uint4 *>(                                              \
          &smem2[BUF][1][0][0] + (size_t)g * 8) = uint4{0, 0, 0, 0};           \
  }

#define GLDS2_MFMA(BUF)                                                        \
  _Pragma("unroll") for (int kk = 0; kk < GB_K; kk += 32) {                    \
    _Pragma("unroll") for (int tj = 0; tj < 4; ++tj) {                         \
      int rb = wc + tj * 16 + li;                                              \
      int cbx = (kk + ke) ^ ((rb & 7) << 3);                                   \
      bf16x8 b0 = *reinterpret_cast<const bf16x8 *>(&smem2[BUF][1][rb][cbx]);  \
      _Pragma("unroll") for (int ti = 0; ti < 8; ++ti) {                       \
        int ra = wr + ti * 16 + li;                                            \
        int ca = (kk + ke) ^ ((ra & 7) << 3);                                  \
        bf16x8 a0 = *reinterpret_cast<const bf16x8 *>(&smem2[BUF][0][ra][ca]); \
        acc[ti][tj] =                                                          \
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[ti][tj],       \
                                                    0, 0, 0);                  \
      }                                                                        \
    }                                                                          \
  }

  GLDS2_STAGE(0, 0)
  __syncthreads();
  int cur = 0;
  for (int k0 = GB_K; k0 < K; k0 += GB_K) {
    GLDS2_STAGE(cur ^ 1, k0)
    GLDS2_MFMA(cur)
    __syncthreads();
    cur ^= 1;
  }
  GLDS2_MFMA(cur)

  int rrow = (lane >> 4) * 4;
#pragma unroll
  for (int ti = 0; ti < 8; ++ti) {
#pragma unroll
    for (int tj = 0; tj < 4; ++tj) {
#pragma unroll
      for (int rg = 0; rg < 4; ++rg) {
        int row = bi + wr + ti * 16 + rrow + rg;
        int col = bj + wc + tj * 16 + li;
        if (row < M && col < N)
          C[(size_t)row * ldc + col] =
              gemm_key(acc[ti][tj][rg], row, col, qn, bn, mode);
      }
    }
  }
#undef GLDS2_STAGE
#undef GLDS2_MFMA
}

// ---------------------------------------------------------------------------
// 3-buffer glds ring variant of the 256² bf16 GEMM (cdna_hip_programming
// §5 "Pipelining across barriers"): BK=32, three LDS buffers (96 KB), one
// tile left IN FLIGHT across each barrier with counted s_waitcnt vmcnt —
// raw s_barrier + lgkmcnt(0) instead of __syncthreads(), whose fence
// would emit vmcnt(0) and drain the pipeline. This kernel runs at
// 1 block/CU (the regime where the guide measures the span at +83% over
// serial staging; the 2-buffer BK=64 k_gemm_bf16_256 is the +40% tier).
// Requirements: K % 32 == 0 and NON-DIVERGENT staging — out-of-range
// rows are address-CLAMPED (garbage lands in C rows/cols >= M/N, which
// the epilogue guard discards) so every wave issues exactly 8 glds per
// tile and the vmcnt counts are exact.
// ---------------------------------------------------------------------------

#define GB3_K 32

extern "C" __global__ __launch_bounds__(512) void k_gemm_bf16_256_p3(
    const unsigned short *__restrict__ A, const unsigned short *__restrict__ B,
    float *__restrict__ C, int M, int N, int K, int lda, int ldb, int ldc,
    const float *__restrict__ qn, const float *__restrict__ bn, int mode) {
  __shared__ unsigned short smem3[3][2][GB2_T][GB3_K];
  int bi = blockIdx.y * GB2_T;
  int bj = blockIdx.x * GB2_T;
  int tid = threadIdx.x;
  int lane = tid & 63, w = tid >> 6;
  int wr = (w >> 2) * 128, wc = (w & 3) * 64;
  f32x4 acc[8][4] = {};
  int li = lane & 15;
  int ke = (lane >> 4) * 8;

  // 1024 16-B groups per operand (256 rows x 4), 8 waves x 2 x 64 lanes;
  // 4 glds per thread per operand-pair iteration => 8 per tile
#define GLDS3_STAGE(BUF, K0)                                                   \
  _Pragma("unroll") for (int i = 0; i < 2; ++i) {                              \
    int g = (w * 2 + i) * 64 + lane;                                           \
    int r = g >> 2;                                                            \
    int c8 = (g & 3) * 8;                                                      \
    int c8s = c8 ^ ((r & 3) << 3); /* BK=32 source-side XOR swizzle */         \
    int ra = bi + r < M ? bi + r : M - 1;                                      \
    int rb = bj + r < N ? bj + r : N - 1;                                      \
    unsigned short *ldst =                                                     \
        &smem3[BUF][0][0][0] + ((size_t)(w * 2 + i) * 64) * 8;                 \
    const unsigned short *ga = &A[(size_t)ra * lda + (K0) + c8s];              \
    const unsigned short *gb = &B[(size_t)rb * ldb + (K0) + c8s];              \
    __builtin_amdgcn_global_load_lds(                                          \
        (const __attribute__((address_space(1))) unsigned int *)ga,            \
        (__attribute__((address_space(3))) unsigned int *)ldst, 16, 0, 0);     \
    unsigned short *ldstB =                                                    \
        &smem3[BUF][1][0][0] + ((size_t)(w * 2 + i) * 64) * 8;                 \
    __builtin_amdgcn_global_load_lds(                                          \
        (const __attribute__((address_space(1))) unsigned int *)gb,            \
        (__attribute__((address_space(3))) unsigned int *)ldstB, 16, 0, 0);    \
  }

#define GLDS3_MFMA(BUF)                                                        \
  _Pragma("unroll") for (int tj = 0; tj < 4; ++tj) {                           \
    int rb = wc + tj * 16 + li;                                                \
    int cbx = ke ^ ((rb & 3) << 3);                                            \
    bf16x8 b0 = *reinterpret_cast<const bf16x8 *>(&smem3[BUF][1][rb][cbx]);    \
    _Pragma("unroll") for (int ti = 0; ti < 8; ++ti) {                         \
      int ra = wr + ti * 16 + li;                                              \
      int ca = ke ^ ((ra & 3) << 3);                                           \
      bf16x8 a0 = *reinterpret_cast<const bf16x8 *>(&smem3[BUF][0][ra][ca]);   \
      acc[ti][tj] =                                                            \
          __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[ti][tj],         \
                                                  0, 0, 0);                    \
    }                                                                          \
  }

#define RAW_BARRIER()                                                          \
  do {                                                                         \
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                         \
    __builtin_amdgcn_s_barrier();                                              \
  } while (0)

  const int T = K >> 5;  // K % 32 == 0 (host-guaranteed)
  GLDS3_STAGE(0, 0)
  if (T > 1) GLDS3_STAGE(1, GB3_K)
  // buf0 landed (the 4 glds of buf1 may stay in flight)
  if (T > 1) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  else asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  RAW_BARRIER();
  for (int t = 0; t < T; ++t) {
    if (t + 2 < T) GLDS3_STAGE((t + 2) % 3, (t + 2) * GB3_K)
    GLDS3_MFMA(t % 3)
    if (t + 1 < T) {
      // next buffer landed; the one after (if staged) stays in flight
      if (t + 2 < T) asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      else asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      RAW_BARRIER();
    }
  }

  int rrow = (lane >> 4) * 4;
#pragma unroll
  for (int ti = 0; ti < 8; ++ti) {
#pragma unroll
    for (int tj = 0; tj < 4; ++tj) {
#pragma unroll
      for (int rg = 0; rg < 4; ++rg) {
        int row = bi + wr + ti * 16 + rrow + rg;
        int col = bj + wc + tj * 16 + li;
        if (row < M && col < N)
          C[(size_t)row * ldc + col] =
              gemm_key(acc[ti][tj][rg], row, col, qn, bn, mode);
      }
    }
  }
#undef GLDS3_STAGE
#undef GLDS3_MFMA
#undef RAW_BARRIER
}

// ---------------------------------------------------------------------------

extern "C" __global__ void k_assign_init(float *best_v, int *best_i, long long n) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    best_v[i] = DFANN_FLT_MAX;
    best_i[i] = -1;
  }
}

// block-per-row running argmin: 256 threads reduce a row of the key
// matrix (a serial per-thread variant was the train-time bottleneck at
// nlist=65536: 1 thread x 65536 sequential reads).
// Ties: lowest column; running compare strict < keeps the earlier
// (lower-col) chunk on ties.
extern "C" __global__ __launch_bounds__(256) void k_assign_rowblock(
    const float *__restrict__ keys, long long rows, long long cols,
    long long ldk, int col_base, float *__restrict__ best_v,
    int *__restrict__ best_i) {
  long long r = blockIdx.x;
  if (r >= rows) return;
  const float *kp = keys + r * ldk;
  float bv = DFANN_FLT_MAX;
  int bi = 0x7FFFFFFF;
  // float4 row reads where aligned (1 KB/wave-load; see k_topk_rows_rk).
  // (value, column) tie-break everywhere -> lane assignment irrelevant.
  if ((ldk & 3) == 0) {
    const float4 *kp4 = reinterpret_cast<const float4 *>(kp);
    long long c4n = cols >> 2;
    long long c4 = threadIdx.x;
    for (; c4 < c4n; c4 += blockDim.x) {
      float4 v4 = kp4[c4];
#pragma unroll
      for (int t = 0; t < 4; ++t) {
        float v = t == 0 ? v4.x : t == 1 ? v4.y : t == 2 ? v4.z : v4.w;
        int c = (int)(c4 * 4 + t);
        if (v < bv || (v == bv && c < bi)) { bv = v; bi = c; }
      }
    }
    for (long long c = (c4n << 2) + threadIdx.x; c < cols; c += blockDim.x) {
      float v = kp[c];
      if (v < bv || (v == bv && (int)c < bi)) { bv = v; bi = (int)c; }
    }
  } else {
    for (long long c = threadIdx.x; c < cols; c += blockDim.x) {
      float v = kp[c];
      if (v < bv || (v == bv && (int)c < bi)) { bv = v; bi = (int)c; }
    }
  }
#pragma unroll
  for (int o = 32; o > 0; o >>= 1) {
    float ov = __shfl_down(bv, o, 64);
    int oi = __shfl_down(bi, o, 64);
    if (ov < bv || (ov == bv && oi < bi)) { bv = ov; bi = oi; }
  }
  __shared__ float wv[4];
  __shared__ int wi[4];
  int lane = threadIdx.x & 63, w = threadIdx.x >> 6;
  if (lane == 0) { wv[w] = bv; wi[w] = bi; }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int i = 1; i < 4; ++i)
      if (wv[i] < bv || (wv[i] == bv && wi[i] < bi)) { bv = wv[i]; bi = wi[i]; }
    if (bv < best_v[r]) {
      best_v[r] = bv;
      best_i[r] = col_base + bi;
    }
  }
}

// ---------------------------------------------------------------------------
// IVF list scan. One block per (query, probe). FAM: 0=PQ, 1=IVF-Flat,
// 2=SQ8, 3=SQfp16. IP: minimize-key = -(bias + sum).
// LDS: [fam region (fam_floats)] [SEL].
// cand_d/cand_p: (nq, nprobe, k); pos = CSR row (uint32).
// keys_dev: coarse minimize-keys (IP bias = -key); null for L2.
// ---------------------------------------------------------------------------

// per-row distance (shared by both selection paths); fam = staged LDS
// region (FAM 0: LUT; FAM 2: [target][vmin][scale]; FAM 3: target).
// Loads are issued in 64-byte batches (4x uint4 up front) so the HBM
// latency of a row's chunks overlaps instead of chaining — the
// accumulation ORDER is unchanged (sequential over the reduced axis,
// contract off), so the oracle bit-exactness contract holds.

// (float) cast: identity for f32 LUTs, h->f32 convert for the fp16-LUT
// approximation path (pq_lut_f16) — the accumulation itself stays f32
#define DFANN_PROC16_PQ(WV, G)                                                 \
  if ((G) < m) {                                                               \
    unsigned w0_ = (WV).x, w1_ = (WV).y, w2_ = (WV).z, w3_ = (WV).w;           \
    _Pragma("unroll") for (int b = 0; b < 16; ++b) {                           \
      if ((G) + b < m) {                                                       \
        unsigned word = (b < 4) ? w0_ : (b < 8) ? w1_ : (b < 12) ? w2_ : w3_;  \
        unsigned c = (word >> ((b & 3) * 8)) & 0xFFu;                          \
        acc = acc + (float)lut[((G) + b) * 256 + c];                           \
      }                                                                        \
    }                                                                          \
  }

// SQ8 distance with the decode algebra FOLDED:
// L2:  diff = u[t] - c*v[t],  u = (q-cent-vmin) - 0.5*scale, v = scale
// IP:  acc += u[t] + c*v[t],  u = q*vmin + 0.5*(q*scale),   v = q*scale
// — mathematically the faiss codec, op-for-op mirrored by the oracle
// (oracle/core.py OracleIVFSQ._scan_one) so distances stay bitwise equal.
// v_cvt_f32_ubyteN converts the byte exactly (one instruction).
// (float)((w >> 8n) & 0xff) — LLVM selects v_cvt_f32_ubyteN for this
#define DFANN_CVT_UB(WORD, B) ((float)(((WORD) >> (((B) & 3) * 8)) & 0xFFu))

#define DFANN_PROC16_SQ8(WV, G)                                                \
  if ((G) < d) {                                                               \
    unsigned w0_ = (WV).x, w1_ = (WV).y, w2_ = (WV).z, w3_ = (WV).w;           \
    _Pragma("unroll") for (int b = 0; b < 16; ++b) {                           \
      if ((G) + b < d) {                                                       \
        unsigned word = (b < 4) ? w0_ : (b < 8) ? w1_ : (b < 12) ? w2_ : w3_;  \
        float cf = DFANN_CVT_UB(word, b);                                      \
        int t = (G) + b;                                                       \
        if (IS_IP) {                                                           \
          acc = acc + (ubuf[t] + cf * vbuf[t]);                                \
        } else {                                                               \
          float diff = ubuf[t] - cf * vbuf[t];                                 \
          acc = acc + diff * diff;                                             \
        }                                                                      \
      }                                                                        \
    }                                                                          \
  }

#define DFANN_PROC8_F16(WV, G)                                                 \
  if ((G) < d) {                                                               \
    _Pragma("unroll") for (int b = 0; b < 8; ++b) {                            \
      if ((G) + b < d) {                                                       \
        unsigned word = (b < 2) ? (WV).x : (b < 4) ? (WV).y                    \
                                 : (b < 6) ? (WV).z : (WV).w;                  \
        unsigned hv = (word >> ((b & 1) * 16)) & 0xFFFFu;                      \
        float dec = __half2float(__ushort_as_half((unsigned short)hv));        \
        int t = (G) + b;                                                       \
        if (IS_IP) acc = acc + rbuf[t] * dec;                                  \
        else {                                                                 \
          float diff = rbuf[t] - dec;                                          \
          acc = acc + diff * diff;                                             \
        }                                                                      \
      }                                                                        \
    }                                                                          \
  }

template <int FAM, bool IS_IP, bool L16 = false>
__device__ __forceinline__ float scan_row_dist(const uint8_t *__restrict__ cp,
                                               const float *__restrict__ fam,
                                               int d, int m) {
  float acc = 0.f;
  const uint4 zero4 = {0, 0, 0, 0};
  if (FAM == 0) {
    using LT = typename std::conditional<L16, __half, float>::type;
    const LT *lut = reinterpret_cast<const LT *>(fam);
    for (int g0 = 0; g0 < m; g0 += 64) {
#pragma clang fp contract(off)
      uint4 wa = *reinterpret_cast<const uint4 *>(cp + g0);
      uint4 wb = (g0 + 16 < m) ? *reinterpret_cast<const uint4 *>(cp + g0 + 16) : zero4;
      uint4 wc = (g0 + 32 < m) ? *reinterpret_cast<const uint4 *>(cp + g0 + 32) : zero4;
      uint4 wd4 = (g0 + 48 < m) ? *reinterpret_cast<const uint4 *>(cp + g0 + 48) : zero4;
      DFANN_PROC16_PQ(wa, g0)
      DFANN_PROC16_PQ(wb, g0 + 16)
      DFANN_PROC16_PQ(wc, g0 + 32)
      DFANN_PROC16_PQ(wd4, g0 + 48)
    }
  } else if (FAM == 2) {
    // full-cacheline batches: a row is one 128-B line (stride 128 for
    // d=128); reading it in one 8x-uint4 burst touches the line once —
    // split batches re-fetched it after L1 eviction (229 KB of rows in
    // flight per CU >> 32 KB L1)
    const float *ubuf = fam, *vbuf = fam + d;
    for (int g0 = 0; g0 < d; g0 += 128) {
#pragma clang fp contract(off)
      uint4 wa = *reinterpret_cast<const uint4 *>(cp + g0);
      uint4 wb = (g0 + 16 < d) ? *reinterpret_cast<const uint4 *>(cp + g0 + 16) : zero4;
      uint4 wc = (g0 + 32 < d) ? *reinterpret_cast<const uint4 *>(cp + g0 + 32) : zero4;
      uint4 wd4 = (g0 + 48 < d) ? *reinterpret_cast<const uint4 *>(cp + g0 + 48) : zero4;
      uint4 we = (g0 + 64 < d) ? *reinterpret_cast<const uint4 *>(cp + g0 + 64) : zero4;
      uint4 wf = (g0 + 80 < d) ? *reinterpret_cast<const uint4 *>(cp + g0 + 80) : zero4;
      uint4 wg = (g0 + 96 < d) ? *reinterpret_cast<const uint4 *>(cp + g0 + 96) : zero4;
      uint4 wh = (g0 + 112 < d) ? *reinterpret_cast<const uint4 *>(cp + g0 + 112) : zero4;
      DFANN_PROC16_SQ8(wa, g0)
      DFANN_PROC16_SQ8(wb, g0 + 16)
      DFANN_PROC16_SQ8(wc, g0 + 32)
      DFANN_PROC16_SQ8(wd4, g0 + 48)
      DFANN_PROC16_SQ8(we, g0 + 64)
      DFANN_PROC16_SQ8(wf, g0 + 80)
      DFANN_PROC16_SQ8(wg, g0 + 96)
      DFANN_PROC16_SQ8(wh, g0 + 112)
    }
  } else {  // FAM 3: fp16 codes, 8 dims per 16 B
    const float *rbuf = fam;
    for (int g0 = 0; g0 < d; g0 += 64) {  // 128 B = one line per batch
#pragma clang fp contract(off)
      uint4 wa = *reinterpret_cast<const uint4 *>(cp + (size_t)g0 * 2);
      uint4 wb = (g0 + 8 < d) ? *reinterpret_cast<const uint4 *>(cp + (size_t)(g0 + 8) * 2) : zero4;
      uint4 wc = (g0 + 16 < d) ? *reinterpret_cast<const uint4 *>(cp + (size_t)(g0 + 16) * 2) : zero4;
      uint4 wd4 = (g0 + 24 < d) ? *reinterpret_cast<const uint4 *>(cp + (size_t)(g0 + 24) * 2) : zero4;
      uint4 we = (g0 + 32 < d) ? *reinterpret_cast<const uint4 *>(cp + (size_t)(g0 + 32) * 2) : zero4;
      uint4 wf = (g0 + 40 < d) ? *reinterpret_cast<const uint4 *>(cp + (size_t)(g0 + 40) * 2) : zero4;
      uint4 wg = (g0 + 48 < d) ? *reinterpret_cast<const uint4 *>(cp + (size_t)(g0 + 48) * 2) : zero4;
      uint4 wh = (g0 + 56 < d) ? *reinterpret_cast<const uint4 *>(cp + (size_t)(g0 + 56) * 2) : zero4;
      DFANN_PROC8_F16(wa, g0)
      DFANN_PROC8_F16(wb, g0 + 8)
      DFANN_PROC8_F16(wc, g0 + 16)
      DFANN_PROC8_F16(wd4, g0 + 24)
      DFANN_PROC8_F16(we, g0 + 32)
      DFANN_PROC8_F16(wf, g0 + 40)
      DFANN_PROC8_F16(wg, g0 + 48)
      DFANN_PROC8_F16(wh, g0 + 56)
    }
  }
  return acc;
}

template <int FAM, bool IS_IP, bool REGSEL, bool PRE = false,
          bool GLUT = false, bool L16 = false, bool NT = false, int UR = 4>
__device__ void ivf_scan_body(
    const float *__restrict__ q, const float *__restrict__ cent,
    const float *__restrict__ cb, const float *__restrict__ sq_vmin,
    const float *__restrict__ sq_scale, const int *__restrict__ probes,
    const float *__restrict__ keys, const uint8_t *const *__restrict__ codes,
    const int64_t *__restrict__ off, int nq, int nprobe, int d, int m,
    int dsub, int k, int stride, int rlog, float *__restrict__ cand_d,
    unsigned *__restrict__ cand_p, int fam_floats,
    const float *__restrict__ term2 = nullptr,
    const float *__restrict__ term3 = nullptr,
    const float *__restrict__ qn = nullptr, int fan = 1,
    const float *__restrict__ glut = nullptr) {
  // fan > 1: each (query, probe) list is split into `fan` segments, one
  // block per segment — long-list tail imbalance at low block counts.
  // EXACT results: every segment's top-k contains the segment's global
  // winners; the candidate merge sees (nprobe*fan*k) entries per query.
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float *fam = reinterpret_cast<float *>(smem);
  // REGSEL extraction scratch is used only AFTER the scan loop (behind
  // a __syncthreads), so it ALIASES the fam region — the block's LDS is
  // max(fam, scratch), not the sum (m=64 fp16 LUT: 33.4 -> 32 KB,
  // 4 -> 5 blocks/CU; the scan measured occupancy-proportional). The
  // LDS-buffer selection path uses its region DURING the scan: offset.
  char *selbase = REGSEL ? smem : smem + (size_t)fam_floats * 4;
  long long blk = blockIdx.x;
  int bq = (int)(blk / ((long long)nprobe * fan));
  int rest = (int)(blk % ((long long)nprobe * fan));
  int bp = rest / fan, seg = rest % fan;
  int L = probes[(long long)bq * nprobe + bp];
  long long out_base = (((long long)bq * nprobe + bp) * fan + seg) * k;
  long long l0 = off[L], l1 = off[L + 1];
  long long len = l1 - l0;
  long long s0 = l0 + len * seg / fan;
  long long s1 = l0 + len * (seg + 1) / fan;
  if (s0 == s1) {
    for (int j = threadIdx.x; j < k; j += blockDim.x) {
      cand_d[out_base + j] = DFANN_FLT_MAX;
      cand_p[out_base + j] = PAD_POS;
    }
    return;
  }
  const float *qp = q + (long long)bq * d;
  float bias = 0.f;
  if (IS_IP) bias = -keys[(long long)bq * nprobe + bp];  // q . centroid
  if (PRE)   // full coarse L2 distance: ||q||^2 + (|c|^2 - 2 q.c)
    bias = qn[bq] + keys[(long long)bq * nprobe + bp];

  // --- stage FAM region ---
  if (FAM == 0 && PRE) {
    // LUT from the precomputed tables: 2 row reads instead of the whole
    // codebook; dist = bias + sum_j LUT
    float *lut = fam;
    const float *t2 = term2 + (size_t)L * m * 256;
    const float *t3 = term3 + (size_t)bq * m * 256;
    for (int e = threadIdx.x; e < m * 256; e += blockDim.x)
      lut[e] = t2[e] - 2.0f * t3[e];
  } else if (FAM == 0 && GLUT) {
    // LUT precomputed in HBM by k_pq_lut (one 1-KiB row per (query,
    // probe, subspace)): stage it with coalesced float4 copies instead
    // of recomputing it from the codebook — at m=64 the in-kernel build
    // re-reads the whole 786-KB codebook from L2 per block and is
    // latency-bound (measured: the scan launch runs at ~125 GB/s of
    // algorithmic code bytes at the configs[3] shape). Values are
    // BIT-IDENTICAL to the in-kernel path: k_pq_lut uses the same
    // sequential-t accumulation with contract off.
    // L16: the LUT row is __half (pq_lut_f16 approximation — half the
    // HBM round-trip and half the LDS, 4 blocks/CU at m=64)
    size_t row_elems = (size_t)m * 256;
    uint4 *dst = reinterpret_cast<uint4 *>(fam);
    const uint4 *src = reinterpret_cast<const uint4 *>(
        reinterpret_cast<const char *>(glut) +
        ((size_t)bq * nprobe + bp) * row_elems * (L16 ? 2 : 4));
    int n4 = (int)(row_elems * (L16 ? 2 : 4) / 16);
    // glds: 16-B global->LDS DMA, no VGPR round-trip — the copy is
    // lane-linear (lane i of each pass writes base + i*16), exactly the
    // wave-uniform-base + lane x size form the DMA requires; the
    // staging barrier below drains it (vmcnt(0) under outstanding glds)
    for (int e = threadIdx.x; e < n4; e += blockDim.x) {
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int *)(src + e),
          (__attribute__((address_space(3))) unsigned int *)(dst + e), 16, 0,
          0);
    }
  } else if (FAM == 0) {
    // rbuf (d floats) AFTER the LUT
    float *lut = fam;
    float *rbuf = fam + (size_t)m * 256;
    for (int t = threadIdx.x; t < d; t += blockDim.x)
      rbuf[t] = IS_IP ? qp[t] : qp[t] - cent[(long long)L * d + t];
    __syncthreads();
    for (int e = threadIdx.x; e < m * 256; e += blockDim.x) {
      int j = e >> 8, c = e & 255;
      const float *cbe = cb + ((size_t)j * 256 + c) * dsub;
      const float *rs = rbuf + j * dsub;
      float acc = 0.f;
      if ((dsub & 3) == 0 && dsub <= 32) {
        // float4 codebook loads (up to 8 in flight); accumulation order
        // unchanged (t ascending, mul+add, contract off) — bit-exact
        float4 cv[8];
        int nv = dsub >> 2;
#pragma unroll
        for (int v = 0; v < 8; ++v)
          if (v < nv) cv[v] = reinterpret_cast<const float4 *>(cbe)[v];
#pragma unroll
        for (int v = 0; v < 8; ++v) {
          if (v < nv) {
#pragma clang fp contract(off)
            for (int tt = 0; tt < 4; ++tt) {
              float cbv = tt == 0 ? cv[v].x : tt == 1 ? cv[v].y
                          : tt == 2 ? cv[v].z : cv[v].w;
              int t = v * 4 + tt;
              if (IS_IP) {
                acc = acc + rs[t] * cbv;
              } else {
                float diff = rs[t] - cbv;
                acc = acc + diff * diff;
              }
            }
          }
        }
      } else {
        for (int t = 0; t < dsub; ++t) {
#pragma clang fp contract(off)
          if (IS_IP) {
            acc = acc + rs[t] * cbe[t];
          } else {
            float diff = rs[t] - cbe[t];
            acc = acc + diff * diff;
          }
        }
      }
      lut[e] = acc;
    }
  } else if (FAM == 1) {
    for (int t = threadIdx.x; t < d; t += blockDim.x) fam[t] = qp[t];
  } else if (FAM == 2) {
    // folded SQ8 terms (see DFANN_PROC16_SQ8); exact op order mirrored in
    // the oracle: 0.5f*scale is an exact exponent decrement
    float *ubuf = fam, *vbuf = fam + d;
    for (int t = threadIdx.x; t < d; t += blockDim.x) {
      float sct = sq_scale[t];
      if (IS_IP) {
        float qsc = qp[t] * sct;
        vbuf[t] = qsc;
        ubuf[t] = qp[t] * sq_vmin[t] + 0.5f * qsc;
      } else {
        float r = qp[t] - cent[(long long)L * d + t];
        ubuf[t] = (r - sq_vmin[t]) - 0.5f * sct;
        vbuf[t] = sct;
      }
    }
  } else {  // SQfp16
    for (int t = threadIdx.x; t < d; t += blockDim.x)
      fam[t] = IS_IP ? qp[t] : qp[t] - cent[(long long)L * d + t];
  }

  RegTopK<16> loc;
  Sel s;
  if (REGSEL) {
    loc.init();
  } else {
    s = sel_carve(selbase);
    sel_init(s);
  }
  __syncthreads();

  // --- scan ---
  if (FAM == 1) {
    // 16-lane-per-vector tree reduction (tolerance parity path)
    int sub = threadIdx.x & 15, grp = threadIdx.x >> 4;
    const int NG16 = blockDim.x >> 4;  // row groups per block
    for (long long base = s0; base < s1; base += (long long)NG16 * 8) {
      if (!REGSEL) sel_guard(s, k, NG16 * 8);
      for (int u = 0; u < 8; ++u) {
        long long pos = base + (long long)u * NG16 + grp;
        float dist = 0.f;
        bool valid = pos < s1;
        if (valid) {
          const float *vp = reinterpret_cast<const float *>(slab_row(codes, rlog, pos, stride));
          float part = 0.f;
          for (int t = sub; t < d; t += 16) {
#pragma clang fp contract(off)
            if (IS_IP) part = part + fam[t] * vp[t];
            else {
              float diff = fam[t] - vp[t];
              part = part + diff * diff;
            }
          }
#pragma unroll
          for (int o = 8; o > 0; o >>= 1) part += __shfl_xor(part, o, 16);
          dist = IS_IP ? -part : part;
        }
        if (valid && sub == 0) {
          if (REGSEL) loc.push(dist, (unsigned)pos);
          else sel_try(s, dist, (unsigned)pos);
        }
      }
    }
  } else if (FAM == 2) {
    // SQ8: 8 lanes per row — a wave reads 8 consecutive rows as 1 KiB of
    // CONTIGUOUS lines (the streaming-copy access pattern) instead of 64
    // scattered line fragments. Reduction: fixed 3-step butterfly over the
    // 8 lanes, mirrored op-for-op by the oracle (OracleIVFSQ._scan_one) —
    // all lanes converge to the same bitwise sum.
    {
      const float *ubuf = fam, *vbuf = fam + d;
      int g8 = threadIdx.x & 7, grp = threadIdx.x >> 3;  // 32 row-groups
      const bool onechunk = d <= 128;  // lane covers one 16-B chunk
      const int NG8 = blockDim.x >> 3;  // row groups per block
      for (long long base = s0; base < s1; base += (long long)NG8 * UR) {
        if (!REGSEL) sel_guard(s, k, NG8 * UR);
        if (onechunk) {
          // issue all UR row-set loads up front: UR independent HBM
          // requests in flight per wave instead of 1 (latency hiding)
          uint4 wv4[UR];
          bool val4[UR];
#pragma unroll
          for (int u = 0; u < UR; ++u) {
            long long pos = base + (long long)u * NG8 + grp;
            val4[u] = pos < s1;
            int t0 = g8 * 16;
            // NT: non-temporal (evict-first) loads — the code stream is
            // read exactly once per step, keep it out of L2's way
            wv4[u] = (val4[u] && t0 < d)
                         ? (NT ? nt_load16(
                                     slab_row(codes, rlog, pos, stride) + t0)
                               : *reinterpret_cast<const uint4 *>(
                                     slab_row(codes, rlog, pos, stride) + t0))
                         : uint4{0, 0, 0, 0};
          }
#pragma unroll
          for (int u = 0; u < UR; ++u) {
            long long pos = base + (long long)u * NG8 + grp;
            float part = 0.f;
            int t0 = g8 * 16;
            if (t0 < d) {
#pragma clang fp contract(off)
              unsigned w0_ = wv4[u].x, w1_ = wv4[u].y, w2_ = wv4[u].z,
                       w3_ = wv4[u].w;
#pragma unroll
              for (int b = 0; b < 16; ++b) {
                if (t0 + b < d) {
                  unsigned word = (b < 4) ? w0_ : (b < 8) ? w1_ : (b < 12) ? w2_ : w3_;
                  float cf = DFANN_CVT_UB(word, b);
                  int t = t0 + b;
                  if (IS_IP) {
                    part = part + (ubuf[t] + cf * vbuf[t]);
                  } else {
                    float diff = ubuf[t] - cf * vbuf[t];
                    part = part + diff * diff;
                  }
                }
              }
            }
            part += __shfl_xor(part, 4, 8);
            part += __shfl_xor(part, 2, 8);
            part += __shfl_xor(part, 1, 8);
            float dist = IS_IP ? -(bias + part) : part;
            if (val4[u] && g8 == 0) {
              if (REGSEL) loc.push(dist, (unsigned)pos);
              else sel_try(s, dist, (unsigned)pos);
            }
          }
        } else {
#pragma unroll
          for (int u = 0; u < UR; ++u) {
            long long pos = base + (long long)u * NG8 + grp;
            bool valid = pos < s1;
            float part = 0.f;
            if (valid) {
              const uint8_t *cp = slab_row(codes, rlog, pos, stride);
              for (int t0 = g8 * 16; t0 < d; t0 += 128) {
#pragma clang fp contract(off)
                uint4 wv = *reinterpret_cast<const uint4 *>(cp + t0);
                unsigned w0_ = wv.x, w1_ = wv.y, w2_ = wv.z, w3_ = wv.w;
#pragma unroll
                for (int b = 0; b < 16; ++b) {
                  if (t0 + b < d) {
                    unsigned word = (b < 4) ? w0_ : (b < 8) ? w1_ : (b < 12) ? w2_ : w3_;
                    float cf = DFANN_CVT_UB(word, b);
                    int t = t0 + b;
                    if (IS_IP) {
                      part = part + (ubuf[t] + cf * vbuf[t]);
                    } else {
                      float diff = ubuf[t] - cf * vbuf[t];
                      part = part + diff * diff;
                    }
                  }
                }
              }
            }
            part += __shfl_xor(part, 4, 8);
            part += __shfl_xor(part, 2, 8);
            part += __shfl_xor(part, 1, 8);
            float dist = IS_IP ? -(bias + part) : part;
            if (valid && g8 == 0) {
              if (REGSEL) loc.push(dist, (unsigned)pos);
              else sel_try(s, dist, (unsigned)pos);
            }
          }
        }
      }
    }
  } else if (REGSEL) {
    // independent rows per thread per iteration: their load batches
    // overlap, hiding HBM/L2 latency without any block synchronization.
    // FAM 0 rows are 16-64 B (4 in flight, cheap); FAM 2/3 rows are a
    // full cache line each (8 uint4 in registers), so 2 in flight keeps
    // VGPRs ~100 (5 waves/SIMD) instead of 161 (3 waves).
    const int UROWS = (FAM == 0) ? 4 : 2;
    const int BS = blockDim.x;
    for (long long base = s0; base < s1; base += (long long)UROWS * BS) {
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        if (u >= UROWS) break;
        long long pos = base + (long long)u * BS + threadIdx.x;
        if (pos < s1) {
          const uint8_t *cp = slab_row(codes, rlog, pos, stride);
          float acc = scan_row_dist<FAM, IS_IP, L16>(cp, fam, d, m);
          float dist = IS_IP ? -(bias + acc) : (PRE ? bias + acc : acc);
          loc.push(dist, (unsigned)pos);
        }
      }
    }
  } else {
    const int BS = blockDim.x;
    for (long long base = s0; base < s1; base += (long long)2 * BS) {
      sel_guard(s, k, 2 * BS);
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        long long pos = base + (long long)u * BS + threadIdx.x;
        if (pos < s1) {
          const uint8_t *cp = slab_row(codes, rlog, pos, stride);
          float acc = scan_row_dist<FAM, IS_IP, L16>(cp, fam, d, m);
          float dist = IS_IP ? -(bias + acc) : (PRE ? bias + acc : acc);
          sel_try(s, dist, (unsigned)pos);
        }
      }
    }
  }
  __syncthreads();
  if (REGSEL) {
    regtopk_block_extract<16>(loc, k, selbase, cand_d + out_base,
                              cand_p + out_base);
  } else {
    sel_compact(s, k);
    int cnt = *s.cnt;
    for (int j = threadIdx.x; j < k; j += blockDim.x) {
      bool v = j < cnt;
      cand_d[out_base + j] = v ? s.d[j] : DFANN_FLT_MAX;
      cand_p[out_base + j] = v ? s.p[j] : PAD_POS;
    }
  }
}

#define INSTANTIATE_SCAN(NAME, FAM, IS_IP, REGSEL)                             \
  extern "C" __global__ __launch_bounds__(512) void NAME(                      \
      const float *q, const float *cent, const float *cb,                      \
      const float *sq_vmin, const float *sq_scale, const int *probes,          \
      const float *keys, const uint8_t *const *codes, const int64_t *off,      \
      int nq,     \
      int nprobe, int d, int m, int dsub, int k, int stride, int rlog,          \
      float *cand_d, unsigned *cand_p, int fam_floats, int fan) {                             \
    ivf_scan_body<FAM, IS_IP, REGSEL>(q, cent, cb, sq_vmin, sq_scale, probes,  \
                                      keys, codes, off, nq, nprobe, d, m,      \
                                      dsub, k, stride, rlog, cand_d, cand_p,   \
                                      fam_floats, nullptr, nullptr, nullptr,   \
                                      fan);                                    \
  }

#define INSTANTIATE_SCAN_PRE(NAME, REGSEL)                                     \
  extern "C" __global__ __launch_bounds__(512) void NAME(                      \
      const float *q, const float *cent, const float *cb,                      \
      const float *sq_vmin, const float *sq_scale, const int *probes,          \
      const float *keys, const uint8_t *const *codes, const int64_t *off,      \
      int nq,     \
      int nprobe, int d, int m, int dsub, int k, int stride, int rlog,          \
      float *cand_d, unsigned *cand_p, int fam_floats, const float *term2,     \
      const float *term3, const float *qn) {                                   \
    ivf_scan_body<0, false, REGSEL, true>(q, cent, cb, sq_vmin, sq_scale,      \
                                          probes, keys, codes, off, nq,        \
                                          nprobe, d, m, dsub, k, stride, rlog, \
                                          cand_d, cand_p, fam_floats, term2,   \
                                          term3, qn);                          \
  }

INSTANTIATE_SCAN_PRE(k_scan_pq_l2_pre, false)
INSTANTIATE_SCAN_PRE(k_scan_pq_l2_pre_rk, true)

INSTANTIATE_SCAN(k_scan_pq_l2, 0, false, false)
INSTANTIATE_SCAN(k_scan_pq_ip, 0, true, false)
INSTANTIATE_SCAN(k_scan_ivfflat_l2, 1, false, false)
INSTANTIATE_SCAN(k_scan_ivfflat_ip, 1, true, false)
INSTANTIATE_SCAN(k_scan_sq8_l2, 2, false, false)
INSTANTIATE_SCAN(k_scan_sq8_ip, 2, true, false)
INSTANTIATE_SCAN(k_scan_sqf_l2, 3, false, false)
INSTANTIATE_SCAN(k_scan_sqf_ip, 3, true, false)
INSTANTIATE_SCAN(k_scan_pq_l2_rk, 0, false, true)
INSTANTIATE_SCAN(k_scan_pq_ip_rk, 0, true, true)

// GLUT variants: LUT staged from HBM (built by k_pq_lut) instead of
// computed in the scan prologue — engine auto-selects at large m where
// the in-kernel build's per-block codebook re-reads dominate.
#define INSTANTIATE_SCAN_GLUT(NAME, IS_IP, REGSEL)                             \
  extern "C" __global__ __launch_bounds__(512) void NAME(                      \
      const float *q, const float *cent, const float *cb,                      \
      const float *sq_vmin, const float *sq_scale, const int *probes,          \
      const float *keys, const uint8_t *const *codes, const int64_t *off,      \
      int nq,     \
      int nprobe, int d, int m, int dsub, int k, int stride, int rlog,          \
      float *cand_d, unsigned *cand_p, int fam_floats, const float *glut) {                   \
    ivf_scan_body<0, IS_IP, REGSEL, false, true>(                              \
        q, cent, cb, sq_vmin, sq_scale, probes, keys, codes, off, nq, nprobe,  \
        d, m, dsub, k, stride, rlog, cand_d, cand_p, fam_floats, nullptr,      \
        nullptr, nullptr, 1, glut);                                                     \
  }
INSTANTIATE_SCAN_GLUT(k_scan_pq_l2_g, false, false)
INSTANTIATE_SCAN_GLUT(k_scan_pq_ip_g, true, false)
INSTANTIATE_SCAN_GLUT(k_scan_pq_l2_g_rk, false, true)
INSTANTIATE_SCAN_GLUT(k_scan_pq_ip_g_rk, true, true)

// fp16-LUT variants (pq_lut_f16): glut holds __half rows
#define INSTANTIATE_SCAN_GLUT_H(NAME, IS_IP, REGSEL)                           \
  extern "C" __global__ __launch_bounds__(512) void NAME(                      \
      const float *q, const float *cent, const float *cb,                      \
      const float *sq_vmin, const float *sq_scale, const int *probes,          \
      const float *keys, const uint8_t *const *codes, const int64_t *off,      \
      int nq,     \
      int nprobe, int d, int m, int dsub, int k, int stride, int rlog,          \
      float *cand_d, unsigned *cand_p, int fam_floats, const float *glut) {                   \
    ivf_scan_body<0, IS_IP, REGSEL, false, true, true>(                        \
        q, cent, cb, sq_vmin, sq_scale, probes, keys, codes, off, nq, nprobe,  \
        d, m, dsub, k, stride, rlog, cand_d, cand_p, fam_floats, nullptr,      \
        nullptr, nullptr, 1, glut);                                                     \
  }
INSTANTIATE_SCAN_GLUT_H(k_scan_pq_l2_gh, false, false)
INSTANTIATE_SCAN_GLUT_H(k_scan_pq_ip_gh, true, false)
INSTANTIATE_SCAN_GLUT_H(k_scan_pq_l2_gh_rk, false, true)
INSTANTIATE_SCAN_GLUT_H(k_scan_pq_ip_gh_rk, true, true)

// ---------------------------------------------------------------------------
// PERSISTENT fp16-GLUT scan (k <= 16 register selection): a fixed grid of
// blocks loops over (query, probe) pairs with stride gridDim.x, and each
// block PREFETCHES the NEXT pair's ADC table into registers while it
// scans the current pair. Motivation (profiles/r01/pmc_sq_waits_1m.csv +
// r1 ladder): at the headline shape a pair's list is ~190 rows — the
// one-pair-per-block kernel is STAGING-dominated (stage 32 KB LUT, scan
// ~12 KB of codes, die) and phase-converged (53% of wave cycles parked).
// Here the LUT round-trip latency hides under the previous pair's scan +
// top-k extraction, and block phases de-correlate naturally.
//   * values are BIT-IDENTICAL to the one-pair kernel: same LUT bytes,
//     same per-row accumulation order, same (dist, pos) selection.
//   * PFN = LUT uint4 groups per thread (lut_bytes / blockDim / 16);
//     host dispatches among the instantiations below and falls back to
//     the one-pair kernel when the shape doesn't divide.
//   * the RegTopK extraction scratch aliases the LDS LUT (as in
//     ivf_scan_body); the prefetched registers rewrite the full LUT for
//     the next pair after extraction's trailing barrier.
// ---------------------------------------------------------------------------

template <bool IS_IP, int PFN>
__device__ void scan_pq_ghp_body(
    const float *__restrict__ q, const float *__restrict__ cent,
    const float *__restrict__ cb, const float *__restrict__ sq_vmin,
    const float *__restrict__ sq_scale, const int *__restrict__ probes,
    const float *__restrict__ keys, const uint8_t *const *__restrict__ codes,
    const int64_t *__restrict__ off, int nq, int nprobe, int d, int m,
    int dsub, int k, int stride, int rlog, float *__restrict__ cand_d,
    unsigned *__restrict__ cand_p, int fam_floats,
    const float *__restrict__ glut) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float *fam = reinterpret_cast<float *>(smem);
  uint4 *lds4 = reinterpret_cast<uint4 *>(smem);
  const int BS = blockDim.x;
  long long qpn = (long long)nq * nprobe;
  const int lut_u4 = PFN * BS;  // uint4 groups in one pair's fp16 LUT
  const uint4 *lutg = reinterpret_cast<const uint4 *>(glut);

  long long pair = blockIdx.x;
  if (pair >= qpn) return;
  // stage the first pair's LUT via global->LDS DMA
  {
    const uint4 *src = lutg + pair * (size_t)lut_u4;
    for (int e = threadIdx.x; e < lut_u4; e += BS)
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int *)(src + e),
          (__attribute__((address_space(3))) unsigned int *)(lds4 + e), 16, 0,
          0);
  }
  uint4 pf[PFN];
  for (; pair < qpn; pair += gridDim.x) {
    long long nxt = pair + gridDim.x;
    __syncthreads();  // staged/rewritten LUT visible (drains the glds)
    // prefetch the next pair's LUT into registers: these loads fly
    // together with the scan's row loads below
    if (nxt < qpn) {
      const uint4 *src = lutg + nxt * (size_t)lut_u4;
#pragma unroll
      for (int i = 0; i < PFN; ++i) pf[i] = src[i * BS + threadIdx.x];
    }
    // ---- scan this pair (same order/ops as ivf_scan_body REGSEL) ----
    int L = probes[pair];
    long long out_base = pair * k;
    long long s0 = off[L], s1 = off[L + 1];
    float bias = 0.f;
    if (IS_IP) bias = -keys[pair];
    if (s0 == s1) {
      for (int j = threadIdx.x; j < k; j += BS) {
        cand_d[out_base + j] = DFANN_FLT_MAX;
        cand_p[out_base + j] = PAD_POS;
      }
    } else {
      RegTopK<16> loc;
      loc.init();
      for (long long base = s0; base < s1; base += (long long)4 * BS) {
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          long long pos = base + (long long)u * BS + threadIdx.x;
          if (pos < s1) {
            const uint8_t *cp = slab_row(codes, rlog, pos, stride);
            float acc = scan_row_dist<0, IS_IP, true>(cp, fam, d, m);
            float dist = IS_IP ? -(bias + acc) : acc;
            loc.push(dist, (unsigned)pos);
          }
        }
      }
      __syncthreads();  // all lanes done reading the LDS LUT
      regtopk_block_extract<16>(loc, k, smem, cand_d + out_base,
                                cand_p + out_base);
    }
    // install the prefetched LUT (extraction's trailing barrier ensures
    // no lane still reads the old one; empty pairs never read it)
    if (nxt < qpn) {
      __syncthreads();  // pad-write case has no extraction barrier
#pragma unroll
      for (int i = 0; i < PFN; ++i) lds4[i * BS + threadIdx.x] = pf[i];
    }
  }
}

#define INSTANTIATE_SCAN_GHP(NAME, IS_IP, PFN)                                 \
  extern "C" __global__ __launch_bounds__(512) void NAME(                      \
      const float *q, const float *cent, const float *cb,                      \
      const float *sq_vmin, const float *sq_scale, const int *probes,          \
      const float *keys, const uint8_t *const *codes, const int64_t *off,      \
      int nq,     \
      int nprobe, int d, int m, int dsub, int k, int stride, int rlog,          \
      float *cand_d, unsigned *cand_p, int fam_floats, const float *glut) {                   \
    scan_pq_ghp_body<IS_IP, PFN>(q, cent, cb, sq_vmin, sq_scale, probes,       \
                                 keys, codes, off, nq, nprobe, d, m, dsub, k,  \
                                 stride, rlog, cand_d, cand_p, fam_floats,     \
                                 glut);    \
  }

INSTANTIATE_SCAN_GHP(k_scan_pq_l2_ghp2, false, 2)
INSTANTIATE_SCAN_GHP(k_scan_pq_ip_ghp2, true, 2)
INSTANTIATE_SCAN_GHP(k_scan_pq_l2_ghp4, false, 4)
INSTANTIATE_SCAN_GHP(k_scan_pq_ip_ghp4, true, 4)
INSTANTIATE_SCAN_GHP(k_scan_pq_l2_ghp8, false, 8)
INSTANTIATE_SCAN_GHP(k_scan_pq_ip_ghp8, true, 8)
INSTANTIATE_SCAN_GHP(k_scan_pq_l2_ghp16, false, 16)
INSTANTIATE_SCAN_GHP(k_scan_pq_ip_ghp16, true, 16)

// ---------------------------------------------------------------------------
// k_pq_lut: ADC lookup tables to HBM, one 256-entry row per (query,
// probe, subspace) at out[(qp)*m*256 + j*256 + c] — the scan's GLUT
// staging then reads its (query, probe) block as one contiguous,
// perfectly-coalesced m-KiB slab. grid = (qp tiles, m subspaces),
// 256 threads = one code c each. The subspace codebook is staged ONCE
// per block in LDS (odd row stride -> conflict-free reads), so the
// codebook leaves L2 ~QPT times less often than the in-kernel build.
// Residual and accumulation are op-for-op the ones the in-kernel build
// uses (sequential t, mul+add, contract off) -> bit-identical LUTs,
// oracle parity preserved (oracle/core.py adc_scan).
// ---------------------------------------------------------------------------
#define PQ_LUT_QPT 64
// DSUB > 0: compile-time subspace width — the thread's codebook row is
// held in REGISTERS across the row loop (the runtime-dsub build left it
// in LDS and re-read it per row: 12-VGPR kernel, ~2.2 ms/step at the
// configs[3] shape). DSUB == 0: runtime fallback.
// PAIRED (f16 only, DSUB>0): each thread computes TWO adjacent codes
// and stores one __half2 — the 2-B per-lane stores of the plain f16
// variant measured 1.7x slower per byte than 4-B stores (1.3 vs
// 2.3 TB/s); half the wave covers row rr, the other half rr+1, so LDS
// and occupancy are unchanged.
template <typename LUTT, int DSUB, bool PAIRED = false>
__device__ __forceinline__ void pq_lut_body(
    const float *__restrict__ q, const float *__restrict__ cent,
    const float *__restrict__ cb, const int *__restrict__ probes, int nq,
    int nprobe, int d, int m, int dsub_rt, int is_ip,
    LUTT *__restrict__ out) {
  const int dsub = DSUB > 0 ? DSUB : dsub_rt;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  int pad = dsub | 1;  // odd stride -> gcd(pad, banks) == 1
  float *cb_sm = reinterpret_cast<float *>(smem);  // 256 * pad
  float *r_sm = cb_sm + 256 * pad;                 // PQ_LUT_QPT * pad
  int j = blockIdx.y;
  const float *cbj = cb + (size_t)j * 256 * dsub;
  for (int e = threadIdx.x; e < 256 * dsub; e += blockDim.x)
    cb_sm[(e / dsub) * pad + (e % dsub)] = cbj[e];
  long long qpn = (long long)nq * nprobe;
  long long qp0 = (long long)blockIdx.x * PQ_LUT_QPT;
  long long qp1 = qp0 + PQ_LUT_QPT;
  if (qp1 > qpn) qp1 = qpn;
  int nrow = (int)(qp1 - qp0);
  for (int e = threadIdx.x; e < nrow * dsub; e += blockDim.x) {
    int rr = e / dsub, t = e % dsub;
    long long qp = qp0 + rr;
    int bq = (int)(qp / nprobe);
    float qv = q[(size_t)bq * d + (size_t)j * dsub + t];
    if (is_ip) {
      r_sm[rr * pad + t] = qv;
    } else {
      int L = probes[qp];
      r_sm[rr * pad + t] = qv - cent[(size_t)L * d + (size_t)j * dsub + t];
    }
  }
  __syncthreads();
  if (PAIRED && DSUB > 0) {
    // four adjacent codes per thread = four INDEPENDENT accumulation
    // chains (the 2-chain variant measured 63% issue-stall — the
    // contract-off sequential adds are a 4-cycle VALU dependency chain)
    // and one 8-B store; wave w covers row rr4+w. Per-entry op order
    // unchanged (sequential t) — values identical to the scalar build.
    int w4 = threadIdx.x >> 6;            // wave id: row within rr group
    int c0 = (threadIdx.x & 63) * 4;      // four adjacent codes
    float cr[4][DSUB > 0 ? DSUB : 1];
#pragma unroll
    for (int cc = 0; cc < 4; ++cc)
#pragma unroll
      for (int t = 0; t < DSUB; ++t) cr[cc][t] = cb_sm[(c0 + cc) * pad + t];
    for (int rr4 = 0; rr4 < nrow; rr4 += 4) {
      int rr = rr4 + w4;
      if (rr < nrow) {
        const float *rs = r_sm + rr * pad;
        float a[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int t = 0; t < DSUB; ++t) {
#pragma clang fp contract(off)
#pragma unroll
          for (int cc = 0; cc < 4; ++cc) {
            if (is_ip) {
              a[cc] = a[cc] + rs[t] * cr[cc][t];
            } else {
              float dd = rs[t] - cr[cc][t];
              a[cc] = a[cc] + dd * dd;
            }
          }
        }
        __half2 lo, hi;
        lo.x = (__half)a[0];  // round-nearest-even, same values as the
        lo.y = (__half)a[1];  // unpaired variant
        hi.x = (__half)a[2];
        hi.y = (__half)a[3];
        uint2 pk;
        pk.x = *reinterpret_cast<unsigned *>(&lo);
        pk.y = *reinterpret_cast<unsigned *>(&hi);
        reinterpret_cast<uint2 *>(out)[((size_t)(qp0 + rr) * m + j) * 64 +
                                       (threadIdx.x & 63)] = pk;
      }
    }
    return;
  }
  int c = threadIdx.x;
  const float *crow = cb_sm + c * pad;
  if (DSUB > 0) {
    // codebook row in registers (fully unrolled: no dynamic indexing);
    // rs[t] reads are wave-broadcast (all lanes the same address).
    // Same op order (sequential t, mul+add, contract off) — bit-exact.
    float creg[DSUB > 0 ? DSUB : 1];
#pragma unroll
    for (int t = 0; t < DSUB; ++t) creg[t] = crow[t];
    for (int rr = 0; rr < nrow; ++rr) {
      const float *rs = r_sm + rr * pad;
      float acc = 0.f;
#pragma unroll
      for (int t = 0; t < DSUB; ++t) {
#pragma clang fp contract(off)
        if (is_ip) {
          acc = acc + rs[t] * creg[t];
        } else {
          float diff = rs[t] - creg[t];
          acc = acc + diff * diff;
        }
      }
      out[(size_t)(qp0 + rr) * ((size_t)m * 256) + (size_t)j * 256 + c] =
          (LUTT)acc;  // __half: round-nearest-even
    }
  } else {
    for (int rr = 0; rr < nrow; ++rr) {
      const float *rs = r_sm + rr * pad;
      float acc = 0.f;
      for (int t = 0; t < dsub; ++t) {
#pragma clang fp contract(off)
        if (is_ip) {
          acc = acc + rs[t] * crow[t];
        } else {
          float diff = rs[t] - crow[t];
          acc = acc + diff * diff;
        }
      }
      out[(size_t)(qp0 + rr) * ((size_t)m * 256) + (size_t)j * 256 + c] =
          (LUTT)acc;  // __half: round-nearest-even
    }
  }
}

#define DFANN_PQ_LUT_DISPATCH(LUTT, PAIRED, OUT)                               \
  switch (dsub) {                                                              \
    case 2: pq_lut_body<LUTT, 2, PAIRED>(q, cent, cb, probes, nq, nprobe, d,   \
                                         m, dsub, is_ip, OUT); break;          \
    case 4: pq_lut_body<LUTT, 4, PAIRED>(q, cent, cb, probes, nq, nprobe, d,   \
                                         m, dsub, is_ip, OUT); break;          \
    case 6: pq_lut_body<LUTT, 6, PAIRED>(q, cent, cb, probes, nq, nprobe, d,   \
                                         m, dsub, is_ip, OUT); break;          \
    case 8: pq_lut_body<LUTT, 8, PAIRED>(q, cent, cb, probes, nq, nprobe, d,   \
                                         m, dsub, is_ip, OUT); break;          \
    case 12: pq_lut_body<LUTT, 12, PAIRED>(q, cent, cb, probes, nq, nprobe,    \
                                           d, m, dsub, is_ip, OUT); break;     \
    case 16: pq_lut_body<LUTT, 16, PAIRED>(q, cent, cb, probes, nq, nprobe,    \
                                           d, m, dsub, is_ip, OUT); break;     \
    default: pq_lut_body<LUTT, 0, false>(q, cent, cb, probes, nq, nprobe, d,   \
                                         m, dsub, is_ip, OUT); break;          \
  }

extern "C" __global__ __launch_bounds__(256) void k_pq_lut(
    const float *q, const float *cent, const float *cb, const int *probes,
    int nq, int nprobe, int d, int m, int dsub, int is_ip, float *out) {
  DFANN_PQ_LUT_DISPATCH(float, false, out)
}

extern "C" __global__ __launch_bounds__(256) void k_pq_lut_f16(
    const float *q, const float *cent, const float *cb, const int *probes,
    int nq, int nprobe, int d, int m, int dsub, int is_ip, __half *out) {
  DFANN_PQ_LUT_DISPATCH(__half, true, out)
}
#define INSTANTIATE_SCAN_NT(NAME, FAM, IS_IP, REGSEL)                          \
  extern "C" __global__ __launch_bounds__(512) void NAME(                      \
      const float *q, const float *cent, const float *cb,                      \
      const float *sq_vmin, const float *sq_scale, const int *probes,          \
      const float *keys, const uint8_t *const *codes, const int64_t *off,      \
      int nq,                                                                  \
      int nprobe, int d, int m, int dsub, int k, int stride, int rlog,         \
      float *cand_d, unsigned *cand_p, int fam_floats, int fan) {              \
    ivf_scan_body<FAM, IS_IP, REGSEL, false, false, false, true>(              \
        q, cent, cb, sq_vmin, sq_scale, probes, keys, codes, off, nq, nprobe,  \
        d, m, dsub, k, stride, rlog, cand_d, cand_p, fam_floats, nullptr,      \
        nullptr, nullptr, fan);                                                \
  }
INSTANTIATE_SCAN_NT(k_scan_sq8_l2_rk_nt, 2, false, true)
INSTANTIATE_SCAN_NT(k_scan_sq8_ip_rk_nt, 2, true, true)

#define INSTANTIATE_SCAN_U8(NAME, FAM, IS_IP, REGSEL)                          \
  extern "C" __global__ __launch_bounds__(512) void NAME(                      \
      const float *q, const float *cent, const float *cb,                      \
      const float *sq_vmin, const float *sq_scale, const int *probes,          \
      const float *keys, const uint8_t *const *codes, const int64_t *off,      \
      int nq,                                                                  \
      int nprobe, int d, int m, int dsub, int k, int stride, int rlog,         \
      float *cand_d, unsigned *cand_p, int fam_floats, int fan) {              \
    ivf_scan_body<FAM, IS_IP, REGSEL, false, false, false, false, 8>(          \
        q, cent, cb, sq_vmin, sq_scale, probes, keys, codes, off, nq, nprobe,  \
        d, m, dsub, k, stride, rlog, cand_d, cand_p, fam_floats, nullptr,      \
        nullptr, nullptr, fan);                                                \
  }
INSTANTIATE_SCAN_U8(k_scan_sq8_l2_rk_u8, 2, false, true)
INSTANTIATE_SCAN_U8(k_scan_sq8_ip_rk_u8, 2, true, true)

INSTANTIATE_SCAN(k_scan_ivfflat_l2_rk, 1, false, true)
INSTANTIATE_SCAN(k_scan_ivfflat_ip_rk, 1, true, true)
INSTANTIATE_SCAN(k_scan_sq8_l2_rk, 2, false, true)
INSTANTIATE_SCAN(k_scan_sq8_ip_rk, 2, true, true)
INSTANTIATE_SCAN(k_scan_sqf_l2_rk, 3, false, true)
INSTANTIATE_SCAN(k_scan_sqf_ip_rk, 3, true, true)

// ---------------------------------------------------------------------------
// merge scan candidates -> final (D, I) per query.
// cand entries: (minimize-key, pos); pos PAD_POS = skip.
// ids: CSR ids array (pos -> arrival id); tie-break on arrival id.
// If ids == nullptr, pos IS the id (flat chunk winners).
// Output: faiss conventions (IP distances un-negated, pads).
// ---------------------------------------------------------------------------

extern "C" __global__ __launch_bounds__(256) void k_merge_cand(
    const float *__restrict__ cand_d, const unsigned *__restrict__ cand_p,
    long long nq, int C, int k, const int64_t *__restrict__ ids, int is_ip,
    float *__restrict__ D, int64_t *__restrict__ I) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  Sel s = sel_carve(smem);
  long long qi = blockIdx.x;
  if (qi >= nq) return;
  sel_init(s);
  __syncthreads();
  const float *cd = cand_d + qi * C;
  const unsigned *cp = cand_p + qi * C;
  const int BSm = blockDim.x;
  for (int c0 = 0; c0 < C; c0 += 2 * BSm) {
    sel_guard(s, k, 2 * BSm);
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      int c = c0 + u * BSm + threadIdx.x;
      if (c < C) {
        unsigned pos = cp[c];
        if (pos != PAD_POS) {
          unsigned idu = ids ? (unsigned)ids[pos] : pos;
          sel_try(s, cd[c], idu);
        }
      }
    }
  }
  __syncthreads();
  sel_compact(s, k);
  int cnt = *s.cnt;
  for (int j = threadIdx.x; j < k; j += blockDim.x) {
    bool v = j < cnt;
    if (v) {
      D[qi * k + j] = is_ip ? -s.d[j] : s.d[j];
      I[qi * k + j] = (long long)s.p[j];
    } else {
      D[qi * k + j] = is_ip ? -DFANN_FLT_MAX : DFANN_FLT_MAX;
      I[qi * k + j] = -1;
    }
  }
}

// ---------------------------------------------------------------------------
// shard-result merge (client-side heap semantics, ref client.py:265-310):
// inputs (S, nq, k) faiss-convention D + ids; maximize negates on the way
// in and the OUTPUT KEEPS the negation (reference quirk 2). Output ids =
// slot s*nq*k + q*k + j (caller maps to shard metadata). Every input slot
// is a candidate (pads carry FLT_MAX and can win, as in the reference).
// Tie-break: ascending slot (= shard order, then rank).
// ---------------------------------------------------------------------------

extern "C" __global__ __launch_bounds__(256) void k_merge_shards(
    const float *__restrict__ Dall, long long nq, int S, int k, int maximize,
    float *__restrict__ Dout, int64_t *__restrict__ Iout) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  Sel s = sel_carve(smem);
  long long qi = blockIdx.x;
  if (qi >= nq) return;
  sel_init(s);
  __syncthreads();
  int C = S * k;
  const int BSm = blockDim.x;
  for (int c0 = 0; c0 < C; c0 += 2 * BSm) {
    sel_guard(s, k, 2 * BSm);
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      int c = c0 + u * BSm + threadIdx.x;
      if (c < C) {
        int sh = c / k, j = c % k;
        float v = Dall[((long long)sh * nq + qi) * k + j];
        float key = maximize ? -v : v;
        unsigned slot = (unsigned)(sh * k + j);  // dense per-query slot
        sel_try(s, key, slot);
      }
    }
  }
  __syncthreads();
  sel_compact(s, k);
  for (int j = threadIdx.x; j < k; j += blockDim.x) {
    unsigned slot = s.p[j];
    int sh = slot / k, jj = slot % k;
    Dout[qi * k + j] = s.d[j];  // negated for maximize — quirk 2 kept
    Iout[qi * k + j] = ((long long)sh * nq + qi) * k + jj;  // global slot
  }
}

// register-path candidate merge (k <= 16)
extern "C" __global__ __launch_bounds__(256) void k_merge_cand_rk(
    const float *__restrict__ cand_d, const unsigned *__restrict__ cand_p,
    long long nq, int C, int k, const int64_t *__restrict__ ids, int is_ip,
    float *__restrict__ D, int64_t *__restrict__ I) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  long long qi = blockIdx.x;
  if (qi >= nq) return;
  RegTopK<16> loc;
  loc.init();
  const float *cd = cand_d + qi * C;
  const unsigned *cp = cand_p + qi * C;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    unsigned pos = cp[c];
    if (pos != PAD_POS) {
      unsigned idu = ids ? (unsigned)ids[pos] : pos;
      loc.push(cd[c], idu);
    }
  }
  __syncthreads();
  float *od = reinterpret_cast<float *>(smem + REGSEL_LDS_BYTES);
  unsigned *op = reinterpret_cast<unsigned *>(smem + REGSEL_LDS_BYTES + 16 * 4);
  regtopk_block_extract<16>(loc, k, smem, od, op);
  for (int j = threadIdx.x; j < k; j += blockDim.x) {
    bool v = op[j] != PAD_POS;
    if (v) {
      D[qi * k + j] = is_ip ? -od[j] : od[j];
      I[qi * k + j] = (long long)op[j];
    } else {
      D[qi * k + j] = is_ip ? -DFANN_FLT_MAX : DFANN_FLT_MAX;
      I[qi * k + j] = -1;
    }
  }
}

// register-path shard merge (k <= 16)
extern "C" __global__ __launch_bounds__(256) void k_merge_shards_rk(
    const float *__restrict__ Dall, long long nq, int S, int k, int maximize,
    float *__restrict__ Dout, int64_t *__restrict__ Iout) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  long long qi = blockIdx.x;
  if (qi >= nq) return;
  RegTopK<16> loc;
  loc.init();
  int C = S * k;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    int sh = c / k, j = c % k;
    float v = Dall[((long long)sh * nq + qi) * k + j];
    loc.push(maximize ? -v : v, (unsigned)(sh * k + j));
  }
  __syncthreads();
  float *od = reinterpret_cast<float *>(smem + REGSEL_LDS_BYTES);
  unsigned *op = reinterpret_cast<unsigned *>(smem + REGSEL_LDS_BYTES + 16 * 4);
  regtopk_block_extract<16>(loc, k, smem, od, op);
  for (int j = threadIdx.x; j < k; j += blockDim.x) {
    unsigned slot = op[j];
    int sh = slot / k, jj = slot % k;
    Dout[qi * k + j] = od[j];
    Iout[qi * k + j] = ((long long)sh * nq + qi) * k + jj;
  }
}

// ---------------------------------------------------------------------------
// encode / build kernels
// ---------------------------------------------------------------------------

// residual: out[i] = x[i] - cent[assign[i]]
extern "C" __global__ void k_residual(const float *__restrict__ x,
                                      const float *__restrict__ cent,
                                      const int *__restrict__ assign,
                                      long long n, int d,
                                      float *__restrict__ out) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = n * d;
  for (; i < total; i += (long long)gridDim.x * blockDim.x) {
    long long r = i / d, t = i % d;
    out[i] = x[i] - cent[(long long)assign[r] * d + t];
  }
}

// write subspace argmin results into the packed code column j (PQ
// encode = per-subspace distance GEMM + k_assign_rowblock + this);
// writes rows [row0, row0+n) of the slab arena
extern "C" __global__ void k_codes_from_best(const int *__restrict__ best,
                                             long long n, int j, int stride,
                                             int rlog, long long row0,
                                             uint8_t *const *__restrict__ codes) {
  long long p = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  for (; p < n; p += (long long)gridDim.x * blockDim.x)
    slab_row_mut(codes, rlog, row0 + p, stride)[j] = (uint8_t)best[p];
}

// SQ encode (8bit / fp16) of residuals
extern "C" __global__ void k_sq_encode(const float *__restrict__ resid,
                                       const float *__restrict__ vmin,
                                       const float *__restrict__ vdiff,
                                       long long n, int d, int stride,
                                       int is_fp16, int rlog, long long row0,
                                       uint8_t *const *__restrict__ codes) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = n * d;
  for (; i < total; i += (long long)gridDim.x * blockDim.x) {
    long long r = i / d;
    int t = (int)(i % d);
    float v = resid[i];
    uint8_t *row = slab_row_mut(codes, rlog, row0 + r, stride);
    if (is_fp16) {
      __half h = __float2half(v);
      reinterpret_cast<unsigned short *>(row)[t] = __half_as_ushort(h);
    } else {
      float xi = (v - vmin[t]) / vdiff[t];
      int c = (int)(255.0f * xi);  // trunc toward zero, as the oracle
      c = c < 0 ? 0 : (c > 255 ? 255 : c);
      row[t] = (uint8_t)c;
    }
  }
}

// PQ-L2 precomputed tables (faiss IndexIVFPQ use_precomputed_table
// restated): dist(q, code | L) = [qn + coarse_key(q,L)] + sum_j LUT[j][k]
// with LUT[j][k] = term2[L][j][k] - 2*term3[q][j][k],
// term2 = ||cb_jk||^2 + 2 c_{L,j}.cb_jk   (per index, nlist*m*256 fp32),
// term3 = q_j . cb_jk                     (per query batch).
// Cuts the scan block's LUT-build traffic from the full m*256*dsub
// codebook (786 KB at m=64,d=768) to 2 table rows (128 KB).
extern "C" __global__ void k_pq_term2(const float *__restrict__ cent,
                                      const float *__restrict__ cb, int nlist,
                                      int m, int dsub,
                                      float *__restrict__ term2) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = (long long)nlist * m * 256;
  for (; i < total; i += (long long)gridDim.x * blockDim.x) {
    int k = (int)(i & 255);
    int j = (int)((i >> 8) % m);
    long long L = i / (256LL * m);
    const float *cbe = cb + ((size_t)j * 256 + k) * dsub;
    const float *cj = cent + L * (size_t)m * dsub + (size_t)j * dsub;
    float nrm = 0.f, dot = 0.f;
    for (int t = 0; t < dsub; ++t) {
      nrm += cbe[t] * cbe[t];
      dot += cj[t] * cbe[t];
    }
    term2[i] = nrm + 2.0f * dot;
  }
}

extern "C" __global__ void k_pq_term3(const float *__restrict__ q,
                                      const float *__restrict__ cb,
                                      long long nq, int m, int dsub,
                                      float *__restrict__ term3) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = nq * m * 256;
  for (; i < total; i += (long long)gridDim.x * blockDim.x) {
    int k = (int)(i & 255);
    int j = (int)((i >> 8) % m);
    long long qi = i / (256LL * m);
    const float *cbe = cb + ((size_t)j * 256 + k) * dsub;
    const float *qs = q + qi * (size_t)m * dsub + (size_t)j * dsub;
    float dot = 0.f;
    for (int t = 0; t < dsub; ++t) dot += qs[t] * cbe[t];
    term3[i] = dot;
  }
}

// pack raw fp32 rows into the pending byte arena (IVF-Flat codes)
extern "C" __global__ void k_pack_rows(const float *__restrict__ x, long long n,
                                       int d, int stride, int rlog,
                                       long long row0,
                                       uint8_t *const *__restrict__ out) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = n * d;
  for (; i < total; i += (long long)gridDim.x * blockDim.x) {
    long long r = i / d;
    int t = (int)(i % d);
    reinterpret_cast<float *>(slab_row_mut(out, rlog, row0 + r, stride))[t] =
        x[i];
  }
}

// two-source CSR rebuild gather (DESIGN.md §2 memory plan): new CSR row j
// of list l is either an OLD CSR row (same list order) or a PENDING row
// (arrival order within the list, via the psrc permutation). Both source
// arenas and the destination are slab-based; the host launches one call
// per new slab in ascending pos order and frees consumed old slabs
// behind the window. ids: old rows keep their ids; pending row p gets
// id_base + p (ids ARE arrival positions). id2pos rewritten for all.
extern "C" __global__ void k_rebuild_gather(
    long long j0, long long j1, const int64_t *__restrict__ new_off,
    const int64_t *__restrict__ old_off, const int64_t *__restrict__ pend_off,
    const unsigned *__restrict__ psrc,
    const uint8_t *const *__restrict__ old_slabs,
    const uint8_t *const *__restrict__ pend_slabs,
    uint8_t *const *__restrict__ new_slabs,
    const int64_t *__restrict__ old_ids, long long id_base,
    int64_t *__restrict__ new_ids, unsigned *__restrict__ id2pos, int nlist,
    int rlog, int stride) {
  long long j = j0 + (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (j >= j1) return;
  // binary search: largest l with new_off[l] <= j
  int lo = 0, hi = nlist;
  while (lo + 1 < hi) {
    int mid = (lo + hi) >> 1;
    if (new_off[mid] <= j) lo = mid;
    else hi = mid;
  }
  long long r = j - new_off[lo];
  long long old_len = old_off[lo + 1] - old_off[lo];
  const uint8_t *sp;
  long long id;
  if (r < old_len) {
    long long src = old_off[lo] + r;
    sp = slab_row(old_slabs, rlog, src, stride);
    id = old_ids[src];
  } else {
    long long pi = pend_off[lo] + (r - old_len);
    unsigned ps = psrc[pi];
    sp = slab_row(pend_slabs, rlog, (long long)ps, stride);
    id = id_base + (long long)ps;
  }
  uint8_t *dp = slab_row_mut(new_slabs, rlog, j, stride);
  for (int t = 0; t < stride / 16; ++t)
    reinterpret_cast<uint4 *>(dp)[t] = reinterpret_cast<const uint4 *>(sp)[t];
  new_ids[j] = id;
  id2pos[id] = (unsigned)j;
}

// gather rows by index (f32): out[i] = in[idx[i]]
extern "C" __global__ void k_gather_rows(const float *__restrict__ in,
                                         const int64_t *__restrict__ idx,
                                         long long n, int d,
                                         float *__restrict__ out) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = n * d;
  for (; i < total; i += (long long)gridDim.x * blockDim.x) {
    long long r = i / d;
    int t = (int)(i % d);
    out[i] = in[idx[r] * d + t];
  }
}

// strided subsample gather: out[i] = in[(i*n_in)/n_out]  (oracle kmeans)
extern "C" __global__ void k_gather_strided(const float *__restrict__ in,
                                            long long n_in, long long n_out,
                                            int d, float *__restrict__ out) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = n_out * d;
  for (; i < total; i += (long long)gridDim.x * blockDim.x) {
    long long r = i / d;
    int t = (int)(i % d);
    long long sr = (r * n_in) / n_out;
    out[i] = in[sr * d + t];
  }
}

// subspace slice: out[i*dsub + t] = in[i*d + j0 + t]
extern "C" __global__ void k_subspace_slice(const float *__restrict__ in,
                                            long long n, int d, int j0,
                                            int dsub, float *__restrict__ out) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = n * dsub;
  for (; i < total; i += (long long)gridDim.x * blockDim.x) {
    long long r = i / dsub;
    int t = (int)(i % dsub);
    out[i] = in[r * d + j0 + t];
  }
}

// k-means accumulation (fp32 atomics; order nondeterminism documented)
extern "C" __global__ void k_centroid_accum(const float *__restrict__ x,
                                            const int *__restrict__ assign,
                                            long long n, int d,
                                            float *__restrict__ sums,
                                            int *__restrict__ counts) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = n * d;
  for (; i < total; i += (long long)gridDim.x * blockDim.x) {
    long long r = i / d;
    int t = (int)(i % d);
    atomicAdd(&sums[(long long)assign[r] * d + t], x[i]);
    if (t == 0) atomicAdd(&counts[assign[r]], 1);
  }
}

extern "C" __global__ void k_centroid_div(float *__restrict__ cent,
                                          const float *__restrict__ sums,
                                          const int *__restrict__ counts,
                                          long long kcent, int d) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = kcent * d;
  for (; i < total; i += (long long)gridDim.x * blockDim.x) {
    long long c = i / d;
    if (counts[c] > 0) cent[i] = sums[i] / (float)counts[c];
  }
}

// per-dim min/max via monotone uint encoding (order-independent)
__device__ __forceinline__ unsigned f32_enc(float f) {
  unsigned u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}
__device__ __forceinline__ float f32_dec(unsigned u) {
  unsigned v = (u & 0x80000000u) ? (u & 0x7FFFFFFFu) : ~u;
  return __uint_as_float(v);
}

extern "C" __global__ void k_minmax_init(unsigned *mn, unsigned *mx, int d) {
  int t = blockIdx.x * blockDim.x + threadIdx.x;
  if (t < d) {
    mn[t] = 0xFFFFFFFFu;
    mx[t] = 0u;
  }
}

extern "C" __global__ void k_minmax_dims(const float *__restrict__ x, long long n,
                                         int d, unsigned *__restrict__ mn,
                                         unsigned *__restrict__ mx) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = n * d;
  for (; i < total; i += (long long)gridDim.x * blockDim.x) {
    int t = (int)(i % d);
    unsigned e = f32_enc(x[i]);
    atomicMin(&mn[t], e);
    atomicMax(&mx[t], e);
  }
}

extern "C" __global__ void k_minmax_decode(const unsigned *mn, const unsigned *mx,
                                           int d, float *vmin, float *vdiff,
                                           float *scale) {
  int t = blockIdx.x * blockDim.x + threadIdx.x;
  if (t < d) {
    float lo = f32_dec(mn[t]), hi = f32_dec(mx[t]);
    float df = hi - lo;
    if (df == 0.f) df = 1.0f;  // degenerate dim guard (oracle parity)
    vmin[t] = lo;
    vdiff[t] = df;
    scale[t] = df / 255.0f;
  }
}

// ---------------------------------------------------------------------------
// reconstruction: decode result ids -> vectors. One block per (q, j).
// type: 0 flat/ivfflat raw (flat_src != null uses flat arena directly by id,
// else CSR codes row), 2 pq, 3 sq8, 4 sqfp16
// ---------------------------------------------------------------------------

extern "C" __global__ __launch_bounds__(64) void k_reconstruct(
    const int64_t *__restrict__ I, long long nq, int k, int type, int d,
    int m, int dsub, int stride, int rlog, const float *__restrict__ flat_src,
    const uint8_t *const *__restrict__ codes,
    const unsigned *__restrict__ id2pos,
    const int64_t *__restrict__ off, int nlist,
    const float *__restrict__ cent, const float *__restrict__ cb,
    const float *__restrict__ vmin, const float *__restrict__ scale,
    float *__restrict__ R) {
  long long e = blockIdx.x;
  if (e >= nq * k) return;
  long long id = I[e];
  float *out = R + e * d;
  if (id < 0) {
    for (int t = threadIdx.x; t < d; t += blockDim.x) out[t] = 0.f;
    return;
  }
  if (flat_src) {
    const float *src = flat_src + id * d;
    for (int t = threadIdx.x; t < d; t += blockDim.x) out[t] = src[t];
    return;
  }
  if (type == 5) {  // hnsw: non-residual SQ8, codes indexed by id
    const uint8_t *cp5 = slab_row(codes, rlog, id, stride);
    for (int t = threadIdx.x; t < d; t += blockDim.x)
      out[t] = vmin[t] + ((float)cp5[t] + 0.5f) * scale[t];
    return;
  }
  unsigned pos = id2pos[id];
  // binary search list containing pos
  int lo = 0, hi = nlist;
  while (lo + 1 < hi) {
    int mid = (lo + hi) >> 1;
    if (off[mid] <= (long long)pos) lo = mid;
    else hi = mid;
  }
  int L = lo;
  const uint8_t *cp = slab_row(codes, rlog, (long long)pos, stride);
  if (type == 0) {  // ivfflat raw
    const float *src = reinterpret_cast<const float *>(cp);
    for (int t = threadIdx.x; t < d; t += blockDim.x) out[t] = src[t];
  } else if (type == 2) {  // pq: centroid + codebook
    for (int t = threadIdx.x; t < d; t += blockDim.x) {
      int j = t / dsub, tt = t % dsub;
      unsigned c = cp[j];
      out[t] = cent[(long long)L * d + t] + cb[((size_t)j * 256 + c) * dsub + tt];
    }
  } else if (type == 3) {  // sq8
    for (int t = threadIdx.x; t < d; t += blockDim.x) {
      unsigned c = cp[t];
      out[t] = cent[(long long)L * d + t] + (vmin[t] + ((float)c + 0.5f) * scale[t]);
    }
  } else {  // sqfp16
    const unsigned short *hp = reinterpret_cast<const unsigned short *>(cp);
    for (int t = threadIdx.x; t < d; t += blockDim.x) {
      out[t] = cent[(long long)L * d + t] + __half2float(__ushort_as_half(hp[t]));
    }
  }
}
