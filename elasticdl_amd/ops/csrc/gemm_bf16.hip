// Fused dense-tower GEMM for MI355X (gfx950, CDNA4).
//
//   C[M,N] = act(A[M,K] @ B[N,K]^T + bias[N])        (bf16 in, bf16 out)
//
// This is the hand-written MFMA path for the Wide&Deep / DeepFM dense
// towers (the reference's towers are stock TF dense layers; here the
// forward matmul + bias + activation are one kernel). Structure follows
// the CDNA4 guide's verified 128x128-tile recipe:
//   - 256 threads = 4 waves (2x2), each wave owns a 64x64 output quadrant
//     as 4x4 fragments of v_mfma_f32_16x16x32_bf16;
//   - K-loop stages A/B tiles (128x64 bf16 each, 16 KiB) into LDS with
//     __builtin_amdgcn_global_load_lds width 16 (wave-uniform LDS base +
//     lane*16, so the LDS layout is linear/unpadded);
//   - XCD-aware bijective blockIdx swizzle for per-XCD L2 locality;
//   - fused epilogue: bias add + activation + bf16 store, masked for
//     arbitrary M/N (K must be a multiple of 64 — the Python wrapper pads).
//
// Requires: K % 64 == 0, A/B 16-byte aligned (torch allocations are).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define BM 128
#define BN 128
#define BK 64
#define GEMM_THREADS 256

// activation selector
#define ACT_NONE 0
#define ACT_RELU 1
#define ACT_SIGMOID 2

__device__ inline float apply_act(float x, int act) {
  switch (act) {
    case ACT_RELU: return x > 0.f ? x : 0.f;
    case ACT_SIGMOID: return 1.f / (1.f + __expf(-x));
    default: return x;
  }
}

// bijective XCD swizzle (guide ERRATA #11 / m204): contiguous grid chunks
// per XCD so neighboring tiles share an L2.
__device__ inline int xcd_swizzle(int bid, int nwg) {
  const int nxcd = 8;
  if (nwg < nxcd) return bid;
  int xcd = bid % nxcd;
  int idx = bid / nxcd;
  int q = nwg / nxcd, r = nwg % nxcd;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

template <int ACT, bool HAS_BIAS>
__global__ __launch_bounds__(GEMM_THREADS, 2) void gemm_bias_act_kernel(
    const __bf16* __restrict__ A,  // [M, K] row-major
    const __bf16* __restrict__ B,  // [N, K] row-major (torch Linear weight)
    const float* __restrict__ bias,  // [N] or nullptr
    __bf16* __restrict__ C,          // [M, N] row-major
    int M, int N, int K) {
  __shared__ __bf16 As[BM * BK];  // As[m][k] linear
  __shared__ __bf16 Bs[BN * BK];  // Bs[n][k] linear

  const int ntile_m = (M + BM - 1) / BM;
  const int ntile_n = (N + BN - 1) / BN;
  int bid = xcd_swizzle(blockIdx.x, ntile_m * ntile_n);
  const int bm = bid / ntile_n;
  const int bn = bid % ntile_n;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;       // 0..3
  const int wm = wave >> 1;        // 0..1
  const int wn = wave & 1;         // 0..1

  // staging geometry: each wave issues 4 global_load_lds of 16 B/lane per
  // operand tile; chunk index ci in [0, 1024) covers 128 rows x 8 chunks.
  // LDS dest byte = ci*16 (wave-uniform base (wave*4+q)*1024 + lane*16).
  const int row_a0 = bm * BM;
  const int row_b0 = bn * BN;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int frag_m = lane & 15;       // row within 16x16 fragment
  const int frag_k = (lane >> 4) * 8; // k offset of this lane's 8 elements

  for (int k0 = 0; k0 < K; k0 += BK) {
    // ---- stage A and B tiles ----
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      int ci = (wave * 4 + q) * 64 + lane;  // 16B chunk index in tile
      int m = ci >> 3;                      // tile row
      int kc = ci & 7;                      // 16B chunk within row
      int ga = min(row_a0 + m, M - 1);
      int gb = min(row_b0 + m, N - 1);
      const __bf16* srcA = A + (size_t)ga * K + k0 + kc * 8;
      const __bf16* srcB = B + (size_t)gb * K + k0 + kc * 8;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)srcA,
          (__attribute__((address_space(3))) void*)(As + (wave * 4 + q) * 512),
          16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)srcB,
          (__attribute__((address_space(3))) void*)(Bs + (wave * 4 + q) * 512),
          16, 0, 0);
    }
    asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();

    // ---- compute: 2 k-chunks of 32, 4x4 fragments ----
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 a_frag[4], b_frag[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int am = wm * 64 + i * 16 + frag_m;
        a_frag[i] = *reinterpret_cast<const bf16x8*>(
            As + am * BK + kk * 32 + frag_k);
        int bnr = wn * 64 + i * 16 + frag_m;
        b_frag[i] = *reinterpret_cast<const bf16x8*>(
            Bs + bnr * BK + kk * 32 + frag_k);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: bias + activation + masked bf16 store ----
  // C/D fragment layout (guide §3, HW-verified): col = lane&15,
  // row = (lane>>4)*4 + reg.
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row_a0 + wm * 64 + i * 16 + (lane >> 4) * 4 + r;
        int col = row_b0 + wn * 64 + j * 16 + (lane & 15);
        if (row < M && col < N) {
          float v = acc[i][j][r];
          if (HAS_BIAS) v += bias[col];
          v = apply_act(v, ACT);
          C[(size_t)row * N + col] = (__bf16)v;
        }
      }
    }
  }
}

// ===========================================================================
// 256x256-tile deep-pipelined kernel (the "phase 2" compute-bound path).
//
// The 128x128 2-barrier structure above tops out near ~900 TF on 4096^3+
// shapes (the guide's measured ceiling for that structure: the vmcnt(0)
// drain before every __syncthreads stalls the staging queue). This kernel
// follows the guide's verified 256^2 8-phase template:
//   - 512 threads = 8 waves (2M x 4N); each wave owns a 128x64 output
//     sub-tile as 8x4 fragments of v_mfma_f32_16x16x32_bf16;
//   - LDS 128 KiB: double-buffered K-tiles (A,B each 256x64 bf16 as two
//     128-row halves), staged with global_load_lds width 16;
//   - st_16x32 LDS swizzle (physical byte = logical ^ ((logical>>9)&1)<<5
//     within each half-tile) kills the 16-way ds_read bank conflict of the
//     linear [128][64] layout (guide: 141x fewer conflicts, +35% at 4k);
//     applied on the store side by pre-swizzling the per-lane GLOBAL source
//     (the LDS destination of global_load_lds is wave-uniform+lane*16 and
//     cannot scatter) and on the read side by XORing the ds_read address;
//   - phase-interleaved K-loop: 4 phases per K-tile, each = a few
//     ds_read_b128 fragment loads + raw s_barrier + lgkmcnt(0) +
//     setprio(1)-wrapped MFMA batch (16) + s_barrier, with the NEXT
//     K-tile's 8 global_load_lds front-loaded into phase 0 and waited only
//     ONCE per K-tile (counted issue distance ~= the whole tile's compute,
//     not the per-tile vmcnt(0)-before-barrier drain of the 128^2 loop);
//   - prefetch always targets the buffer NOT being read, so correctness
//     never depends on barrier placement (the template's 3-half-tile-ahead
//     variant overwrites live slots and is race-prone to reproduce).
// ===========================================================================

#define BM2 256
#define BN2 256
#define BK2 64
#define GEMM2_THREADS 512

// LDS half-tile layout: contiguous 16x32 subtiles (1024 B each), i.e.
// logical byte of element (row, k) in a [128][64] bf16 half-tile is
//   ((row>>4)*2 + (k>>5))*1024 + (row&15)*64 + (k&31)*2
// so a fragment-read row stride is 64 B (16 banks) instead of the
// row-major 128 B (32 banks, which aliases 16 lanes onto 2 bank groups
// and defeats the swizzle). The st_16x32 swizzle then XORs byte-bit5
// with byte-bit9 *within* each subtile (rows 8-15 shift 32 B), spreading
// a 16-lane fragment column read across 8 bank groups (~2-way).
__device__ inline int lds_byte(int row, int k) {
  int lb = (((row >> 4) << 1) + (k >> 5)) * 1024 + (row & 15) * 64 +
           (k & 31) * 2;
  return lb ^ (((lb >> 9) & 1) << 5);
}

// inverse for the store side: which logical (row, k) lives at physical
// 16 B chunk c (the global_load_lds destination is linear in lane)
__device__ inline void lds_chunk_src(int c, int& row, int& k) {
  int lb = (c * 16) ^ ((((c * 16) >> 9) & 1) << 5);
  int subtile = lb >> 10;
  int within = lb & 1023;
  row = ((subtile >> 1) << 4) + ((within >> 6) & 15);
  k = ((subtile & 1) << 5) + ((within & 63) >> 1);
}

// BARS: phase-synchronization structure (A/B-tested on hardware; measured
// 4096^3/8192^3 bf16, ab_gemm1/2):
//   0 = pure free-run: no phase barriers, no explicit waits, compiler
//       schedules counted lgkmcnt itself
//   1 = free-run + per-phase lgkmcnt(0) + setprio around MFMA  [1041/1125 TF]
//   2 = one barrier before each MFMA batch                      [987/1039]
//   3 = the template's two-barrier lockstep                     [859/941]
//   4 = whole-tile fragment reads then whole-tile MFMA (no phases)
//   5 = k-chunk software pipelining with two register sets (spills; slow)
// Production default is 0 — the full hardware A/B ladder is recorded in
// profiles/kernels_r02.md (free-run > +asm > barrier variants > BK=32
// vmcnt ring, on this structure).
template <int ACT, bool HAS_BIAS, int BARS = 0>
__global__ __launch_bounds__(GEMM2_THREADS, 1) void gemm_bias_act_256_kernel(
    const __bf16* __restrict__ A,  // [M, K] row-major
    const __bf16* __restrict__ B,  // [N, K] row-major
    const float* __restrict__ bias,
    __bf16* __restrict__ C,  // [M, N] row-major
    int M, int N, int K) {
  // [dbuf][operand][half][128*64]
  __shared__ __bf16 sm[2][2][2][128 * BK2];

  const int ntile_m = (M + BM2 - 1) / BM2;
  const int ntile_n = (N + BN2 - 1) / BN2;
  int bid = xcd_swizzle(blockIdx.x, ntile_m * ntile_n);
  const int bm = bid / ntile_n;
  const int bn = bid % ntile_n;
  const int row_a0 = bm * BM2;
  const int row_b0 = bn * BN2;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;  // 0..7
  const int wm = wave >> 2;   // 0..1  -> A half = wm
  const int wn = wave & 3;    // 0..3  -> B half = wn>>1

  // ---- staging: one half-tile = 1024 16B chunks = 2 loads/thread. The
  // LDS destination is linear (wave-uniform base + lane*16); the swizzled
  // subtile layout is realized by pre-swizzling the per-lane GLOBAL
  // source: chunk c fetches the (row, k) that lds_byte maps to 16c.
  auto stage_half = [&](int buf, int op, int half, int t_k0) {
    const __bf16* base = (op == 0 ? A : B);
    int rows_total = (op == 0 ? M : N);
    int r0 = (op == 0 ? row_a0 : row_b0) + half * 128;
#pragma unroll
    for (int l = 0; l < 2; ++l) {
      int c = l * 512 + tid;
      int row, k;
      lds_chunk_src(c, row, k);
      int g = min(r0 + row, rows_total - 1);
      const __bf16* src = base + (size_t)g * K + t_k0 + k;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(
              &sm[buf][op][half][0] + (l * 512 + wave * 64) * 8),
          16, 0, 0);
    }
  };
  auto stage_tile = [&](int buf, int t_k0) {
    stage_half(buf, 0, 0, t_k0);
    stage_half(buf, 0, 1, t_k0);
    stage_half(buf, 1, 0, t_k0);
    stage_half(buf, 1, 1, t_k0);
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int frag_m = lane & 15;
  const int frag_k = (lane >> 4) * 8;

  // swizzled ds_read of one bf16x8 fragment (k multiple of 8, within one
  // 32-col subtile -> 16B aligned physical address)
  auto read_frag = [&](const __bf16* halfbuf, int row, int k) -> bf16x8 {
    return *reinterpret_cast<const bf16x8*>(
        reinterpret_cast<const char*>(halfbuf) + lds_byte(row, k));
  };

  const int ktiles = K / BK2;
  // prologue: tile 0 -> buf 0
  stage_tile(0, 0);
  asm volatile("s_waitcnt vmcnt(0)");
  __builtin_amdgcn_s_barrier();

  const int bcol0_w = (wn & 1) * 64;  // cols within this wave's B half

  if (BARS == 5) {
    // software-pipelined at k-chunk granularity: two fragment register
    // sets; the ds_reads for the NEXT 32-deep k-chunk (or the next
    // K-tile's first chunk, after the staging wait) are issued before
    // the current chunk's 32-MFMA batch, so LDS latency and the
    // buffer-swap wait hide under MFMA issue.
    bf16x8 aA[8], bA[4], aB[8], bB[4];
    auto read_set = [&](bf16x8* af, bf16x8* bf, int buf, int kk) {
      const __bf16* Ah = &sm[buf][0][wm][0];
      const __bf16* Bh = &sm[buf][1][wn >> 1][0];
#pragma unroll
      for (int i = 0; i < 8; ++i)
        af[i] = read_frag(Ah, i * 16 + frag_m, kk + frag_k);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        bf[j] = read_frag(Bh, bcol0_w + j * 16 + frag_m, kk + frag_k);
    };
    auto mfma_set = [&](const bf16x8* af, const bf16x8* bf) {
#pragma unroll
      for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
    };
    read_set(aA, bA, 0, 0);
    for (int t = 0; t < ktiles; ++t) {
      const int b = t & 1;
      if (t + 1 < ktiles) stage_tile(b ^ 1, (t + 1) * BK2);
      read_set(aB, bB, b, 32);
      mfma_set(aA, bA);
      if (t + 1 < ktiles) {
        asm volatile("s_waitcnt vmcnt(0)");
        __builtin_amdgcn_s_barrier();
        read_set(aA, bA, b ^ 1, 0);
      }
      mfma_set(aB, bB);
    }
    goto epilogue;
  }

  {
  bf16x8 a_frag[8], b_frag[4];
  for (int t = 0; t < ktiles; ++t) {
    const int b = t & 1;
    const __bf16* Ah = &sm[b][0][wm][0];         // this wave's A half
    const __bf16* Bh0 = &sm[b][1][wn >> 1][0];   // this wave's B half
    const int arow0 = 0;                          // rows within half
    const int bcol0 = bcol0_w;                    // cols within half

    if (BARS == 4) {
      // whole-tile batch: all fragment reads, then all MFMAs; the
      // compiler interleaves with its own counted lgkmcnt waits
      if (t + 1 < ktiles) stage_tile(b ^ 1, (t + 1) * BK2);
      bf16x8 a2[2][8], b2[2][4];
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
        for (int i = 0; i < 8; ++i)
          a2[kk][i] =
              read_frag(Ah, i * 16 + frag_m, kk * 32 + frag_k);
#pragma unroll
        for (int j = 0; j < 4; ++j)
          b2[kk][j] =
              read_frag(Bh0, bcol0 + j * 16 + frag_m, kk * 32 + frag_k);
      }
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int i = 0; i < 8; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a2[kk][i], b2[kk][j], acc[i][j], 0, 0, 0);
    } else {
#pragma unroll
      for (int phase = 0; phase < 4; ++phase) {
        const int kk = (phase >> 1) * 32;  // k-chunk of this phase
        if (phase == 0 && t + 1 < ktiles) {
          // front-load the next tile's 8 staging ops; waited at loop end
          stage_tile(b ^ 1, (t + 1) * BK2);
        }
        if ((phase & 1) == 0) {
#pragma unroll
          for (int i = 0; i < 8; ++i)
            a_frag[i] = read_frag(Ah, arow0 + i * 16 + frag_m, kk + frag_k);
#pragma unroll
          for (int j = 0; j < 2; ++j)
            b_frag[j] = read_frag(Bh0, bcol0 + j * 16 + frag_m, kk + frag_k);
        } else {
#pragma unroll
          for (int j = 2; j < 4; ++j)
            b_frag[j] = read_frag(Bh0, bcol0 + j * 16 + frag_m, kk + frag_k);
        }
        if (BARS >= 2) __builtin_amdgcn_s_barrier();
        if (BARS >= 1) {
          asm volatile("s_waitcnt lgkmcnt(0)");
          __builtin_amdgcn_s_setprio(1);
        }
        const int j0 = (phase & 1) * 2;
#pragma unroll
        for (int i = 0; i < 8; ++i)
#pragma unroll
          for (int j = 0; j < 2; ++j)
            acc[i][j0 + j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a_frag[i], b_frag[j0 + j], acc[i][j0 + j], 0, 0, 0);
        if (BARS >= 1) __builtin_amdgcn_s_setprio(0);
        if (BARS >= 3) __builtin_amdgcn_s_barrier();
      }
    }
    if (t + 1 < ktiles) {
      asm volatile("s_waitcnt vmcnt(0)");
      __builtin_amdgcn_s_barrier();
    }
  }

  }  // non-pipelined variants

epilogue:
  // ---- epilogue (C/D layout: col = lane&15, row = (lane>>4)*4 + reg)
#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row_a0 + wm * 128 + i * 16 + (lane >> 4) * 4 + r;
        int col = row_b0 + wn * 64 + j * 16 + (lane & 15);
        if (row < M && col < N) {
          float v = acc[i][j][r];
          if (HAS_BIAS) v += bias[col];
          v = apply_act(v, ACT);
          C[(size_t)row * N + col] = (__bf16)v;
        }
      }
    }
  }
}

// ===========================================================================
// BK=32 4-slot ring variant: the never-drain vmcnt pipeline. Four 32 KiB
// K-tiles live in LDS; every iteration waits s_waitcnt vmcnt(8) (its own
// tile's 4 staging loads done, the next TWO tiles' 8 loads still in
// flight) — the staging queue is never drained, which is the structural
// fix the guide identifies for the ~900 TF 2-barrier ceiling (AITER's
// interleaved MFMA<->buffer_load with vmcnt(2/5/12), never 0).
// Safety: the slot overwritten by tile t+3's staging is tile t-1's,
// whose ds_reads completed before every wave crossed this iteration's
// barrier (reads are consumed by MFMAs that precede the barrier in
// program order). Requires K >= 4*32 (dispatcher guards).
// ===========================================================================

__device__ inline int lds_byte32(int row, int k) {
  // [128][32] half-tile as contiguous 16x32 subtiles + st_16x32 swizzle
  int lb = ((row >> 4) << 10) + (row & 15) * 64 + k * 2;
  return lb ^ (((lb >> 9) & 1) << 5);
}

template <int ACT, bool HAS_BIAS>
__global__ __launch_bounds__(GEMM2_THREADS, 1) void gemm_bias_act_bk32_kernel(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    const float* __restrict__ bias, __bf16* __restrict__ C, int M, int N,
    int K) {
  __shared__ __bf16 sm[4][2][2][128 * 32];  // [slot][op][half][8 KiB]

  const int ntile_m = (M + BM2 - 1) / BM2;
  const int ntile_n = (N + BN2 - 1) / BN2;
  int bid = xcd_swizzle(blockIdx.x, ntile_m * ntile_n);
  const int bm = bid / ntile_n;
  const int bn = bid % ntile_n;
  const int row_a0 = bm * BM2;
  const int row_b0 = bn * BN2;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave >> 2;
  const int wn = wave & 3;
  const int bcol0 = (wn & 1) * 64;

  // one half-tile (128x32 = 512 chunks) = 1 load/thread
  auto stage_half = [&](int slot, int op, int half, int t_k0) {
    const __bf16* base = (op == 0 ? A : B);
    int rows_total = (op == 0 ? M : N);
    int r0 = (op == 0 ? row_a0 : row_b0) + half * 128;
    int c = tid;
    int lb = (c * 16) ^ ((((c * 16) >> 9) & 1) << 5);
    int row = ((lb >> 10) << 4) + ((lb >> 6) & 15);
    int k = (lb & 63) >> 1;
    int g = min(r0 + row, rows_total - 1);
    const __bf16* src = base + (size_t)g * K + t_k0 + k;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(
            &sm[slot][op][half][0] + wave * 64 * 8),
        16, 0, 0);
  };
  auto stage_tile = [&](int slot, int t_k0) {
    stage_half(slot, 0, 0, t_k0);
    stage_half(slot, 0, 1, t_k0);
    stage_half(slot, 1, 0, t_k0);
    stage_half(slot, 1, 1, t_k0);
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int frag_m = lane & 15;
  const int frag_k = (lane >> 4) * 8;
  auto read_frag32 = [&](const __bf16* halfbuf, int row, int k) -> bf16x8 {
    return *reinterpret_cast<const bf16x8*>(
        reinterpret_cast<const char*>(halfbuf) + lds_byte32(row, k));
  };

  const int ktiles = K / 32;  // caller guarantees >= 4
  stage_tile(0, 0);
  stage_tile(1, 32);
  stage_tile(2, 64);

  bf16x8 a_frag[8], b_frag[4];
  for (int t = 0; t < ktiles; ++t) {
    const int slot = t & 3;
    // own tile complete; the next two tiles' 8 loads stay in flight
    if (t + 2 < ktiles) {
      asm volatile("s_waitcnt vmcnt(8)");
    } else if (t + 1 < ktiles) {
      asm volatile("s_waitcnt vmcnt(4)");
    } else {
      asm volatile("s_waitcnt vmcnt(0)");
    }
    __builtin_amdgcn_s_barrier();
    if (t + 3 < ktiles) stage_tile((t + 3) & 3, (t + 3) * 32);

    const __bf16* Ah = &sm[slot][0][wm][0];
    const __bf16* Bh = &sm[slot][1][wn >> 1][0];
#pragma unroll
    for (int i = 0; i < 8; ++i)
      a_frag[i] = read_frag32(Ah, i * 16 + frag_m, frag_k);
#pragma unroll
    for (int j = 0; j < 4; ++j)
      b_frag[j] = read_frag32(Bh, bcol0 + j * 16 + frag_m, frag_k);
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[i], b_frag[j], acc[i][j], 0, 0, 0);
  }

#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row_a0 + wm * 128 + i * 16 + (lane >> 4) * 4 + r;
        int col = row_b0 + wn * 64 + j * 16 + (lane & 15);
        if (row < M && col < N) {
          float v = acc[i][j][r];
          if (HAS_BIAS) v += bias[col];
          v = apply_act(v, ACT);
          C[(size_t)row * N + col] = (__bf16)v;
        }
      }
    }
  }
}

// benchmark-only entry: pick the barrier variant explicitly
extern "C" void edl_gemm256_variant_bf16(const void* A, const void* B,
                                         void* C, int M, int N, int K,
                                         int bars, hipStream_t stream) {
  int ntiles = ((M + BM2 - 1) / BM2) * ((N + BN2 - 1) / BN2);
  dim3 grid(ntiles), block(GEMM2_THREADS);
  const __bf16* a = reinterpret_cast<const __bf16*>(A);
  const __bf16* b = reinterpret_cast<const __bf16*>(B);
  __bf16* c = reinterpret_cast<__bf16*>(C);
  switch (bars) {
    case 0:
      gemm_bias_act_256_kernel<ACT_NONE, false, 0>
          <<<grid, block, 0, stream>>>(a, b, nullptr, c, M, N, K);
      break;
    case 1:
      gemm_bias_act_256_kernel<ACT_NONE, false, 1>
          <<<grid, block, 0, stream>>>(a, b, nullptr, c, M, N, K);
      break;
    case 2:
      gemm_bias_act_256_kernel<ACT_NONE, false, 2>
          <<<grid, block, 0, stream>>>(a, b, nullptr, c, M, N, K);
      break;
    case 4:
      gemm_bias_act_256_kernel<ACT_NONE, false, 4>
          <<<grid, block, 0, stream>>>(a, b, nullptr, c, M, N, K);
      break;
    case 5:
      gemm_bias_act_256_kernel<ACT_NONE, false, 5>
          <<<grid, block, 0, stream>>>(a, b, nullptr, c, M, N, K);
      break;
    case 6:
      gemm_bias_act_bk32_kernel<ACT_NONE, false>
          <<<grid, block, 0, stream>>>(a, b, nullptr, c, M, N, K);
      break;
    default:
      gemm_bias_act_256_kernel<ACT_NONE, false, 3>
          <<<grid, block, 0, stream>>>(a, b, nullptr, c, M, N, K);
  }
}

extern "C" void edl_gemm_bias_act_bf16(const void* A, const void* B,
                                       const float* bias, void* C, int M,
                                       int N, int K, int act,
                                       hipStream_t stream) {
  const __bf16* a = reinterpret_cast<const __bf16*>(A);
  const __bf16* b = reinterpret_cast<const __bf16*>(B);
  __bf16* c = reinterpret_cast<__bf16*>(C);
  // compute-bound shapes take the 256^2 8-phase kernel; tower shapes keep
  // the 128^2 fused kernel whose smaller tiles fill the grid and win on
  // memory-bound work. The 256^2 kernel runs 1 block/CU (128 KiB LDS), so
  // it also needs >=256 tiles to fill the chip — measured: 4096x1024x2048
  // (64 tiles) is 2.1x FASTER on the 128^2 kernel.
  const long ntiles256 =
      (long)((M + BM2 - 1) / BM2) * ((N + BN2 - 1) / BN2);
  const bool big = (M >= 256) && (N >= 256) && ntiles256 >= 256;
  if (big) {
    int ntiles = ((M + BM2 - 1) / BM2) * ((N + BN2 - 1) / BN2);
    dim3 grid(ntiles), block(GEMM2_THREADS);
#define EDL_GEMM2_CASE(ACTV)                                                  \
  {                                                                           \
    if (bias != nullptr)                                                      \
      gemm_bias_act_256_kernel<ACTV, true>                                    \
          <<<grid, block, 0, stream>>>(a, b, bias, c, M, N, K);               \
    else                                                                      \
      gemm_bias_act_256_kernel<ACTV, false>                                   \
          <<<grid, block, 0, stream>>>(a, b, nullptr, c, M, N, K);            \
  }
    switch (act) {
      case ACT_RELU: EDL_GEMM2_CASE(ACT_RELU); break;
      case ACT_SIGMOID: EDL_GEMM2_CASE(ACT_SIGMOID); break;
      default: EDL_GEMM2_CASE(ACT_NONE); break;
    }
#undef EDL_GEMM2_CASE
    return;
  }
  int ntiles = ((M + BM - 1) / BM) * ((N + BN - 1) / BN);
  dim3 grid(ntiles), block(GEMM_THREADS);
#define EDL_GEMM_CASE(ACTV)                                                   \
  {                                                                           \
    if (bias != nullptr)                                                      \
      gemm_bias_act_kernel<ACTV, true>                                        \
          <<<grid, block, 0, stream>>>(a, b, bias, c, M, N, K);               \
    else                                                                      \
      gemm_bias_act_kernel<ACTV, false>                                       \
          <<<grid, block, 0, stream>>>(a, b, nullptr, c, M, N, K);            \
  }
  switch (act) {
    case ACT_RELU: EDL_GEMM_CASE(ACT_RELU); break;
    case ACT_SIGMOID: EDL_GEMM_CASE(ACT_SIGMOID); break;
    default: EDL_GEMM_CASE(ACT_NONE); break;
  }
#undef EDL_GEMM_CASE
}
