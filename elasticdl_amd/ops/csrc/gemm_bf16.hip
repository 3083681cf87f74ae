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

extern "C" void edl_gemm_bias_act_bf16(const void* A, const void* B,
                                       const float* bias, void* C, int M,
                                       int N, int K, int act,
                                       hipStream_t stream) {
  int ntiles = ((M + BM - 1) / BM) * ((N + BN - 1) / BN);
  dim3 grid(ntiles), block(GEMM_THREADS);
  const __bf16* a = reinterpret_cast<const __bf16*>(A);
  const __bf16* b = reinterpret_cast<const __bf16*>(B);
  __bf16* c = reinterpret_cast<__bf16*>(C);
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
