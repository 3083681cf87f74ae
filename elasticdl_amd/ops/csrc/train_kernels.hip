// Worker-side training kernels for MI355X (gfx950).
//
// The AllReduce data-parallel path keeps the model in bf16 and allreduces
// bf16 gradient buckets over RCCL/xGMI (half the bytes of the reference's
// fp32-per-tensor Horovod path — its own benchmark called out per-tensor
// allreduce + host round-trips as the bottleneck,
// docs/benchmark/ftlib_benchmark.md:176-199). The optimizer step is one
// fused kernel per flat bucket: f32 master weights + f32 momentum are
// updated from the bf16 gradient view and the bf16 model parameter is
// re-materialized in the same pass — 3 reads + 3 writes per element,
// HBM-bound, vectorized 4-wide.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

#define THREADS 256
#define MAX_BLOCKS 2048

typedef __attribute__((ext_vector_type(4))) float f32x4_t;
typedef __attribute__((ext_vector_type(4))) short s16x4_t;

static inline int grid_for(int64_t work_items) {
  int64_t blocks = (work_items + THREADS - 1) / THREADS;
  if (blocks > MAX_BLOCKS) blocks = MAX_BLOCKS;
  if (blocks < 1) blocks = 1;
  return static_cast<int>(blocks);
}

__device__ inline float bf16_bits_to_f32(short bits) {
  union { float f; uint32_t u; } cvt;
  cvt.u = ((uint32_t)(uint16_t)bits) << 16;
  return cvt.f;
}

__device__ inline short f32_to_bf16_bits(float f) {
  union { float f; uint32_t u; } cvt;
  cvt.f = f;
  // round-to-nearest-even
  uint32_t lsb = (cvt.u >> 16) & 1u;
  uint32_t rounded = cvt.u + 0x7fffu + lsb;
  return (short)(rounded >> 16);
}

// p_bf16  : bf16 model parameter (output of the step)
// master  : f32 master weights
// vel     : f32 momentum
// g_bf16  : bf16 gradient (already allreduce-averaged)
__global__ void fused_sgd_bf16_kernel(short* __restrict__ p_bf16,
                                      float* __restrict__ master,
                                      float* __restrict__ vel,
                                      const short* __restrict__ g_bf16,
                                      int64_t numel, float lr, float mu,
                                      bool nesterov, float weight_decay,
                                      float grad_scale) {
  int64_t n4 = numel >> 2;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    s16x4_t gb = reinterpret_cast<const s16x4_t*>(g_bf16)[i];
    f32x4_t m = reinterpret_cast<const f32x4_t*>(master)[i];
    f32x4_t v = reinterpret_cast<const f32x4_t*>(vel)[i];
    s16x4_t pb;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      float g = grad_scale * bf16_bits_to_f32(gb[c]) + weight_decay * m[c];
      float nv = mu * v[c] + g;
      v[c] = nv;
      float upd = nesterov ? (g + mu * nv) : nv;
      m[c] -= lr * upd;
      pb[c] = f32_to_bf16_bits(m[c]);
    }
    reinterpret_cast<f32x4_t*>(master)[i] = m;
    reinterpret_cast<f32x4_t*>(vel)[i] = v;
    reinterpret_cast<s16x4_t*>(p_bf16)[i] = pb;
  }
  if (blockIdx.x == 0) {
    for (int64_t i = (n4 << 2) + threadIdx.x; i < numel; i += blockDim.x) {
      float g = grad_scale * bf16_bits_to_f32(g_bf16[i]) + weight_decay * master[i];
      float nv = mu * vel[i] + g;
      vel[i] = nv;
      master[i] -= lr * (nesterov ? (g + mu * nv) : nv);
      p_bf16[i] = f32_to_bf16_bits(master[i]);
    }
  }
}

// Same fusion for Adam (used by the CTR models' dense side when trained
// allreduce-style, and by anyone calling the worker-side fused AdamW).
__global__ void fused_adamw_bf16_kernel(short* __restrict__ p_bf16,
                                        float* __restrict__ master,
                                        float* __restrict__ m_state,
                                        float* __restrict__ v_state,
                                        const short* __restrict__ g_bf16,
                                        int64_t numel, float lr_t, float b1,
                                        float b2, float eps,
                                        float weight_decay, float lr,
                                        float grad_scale) {
  int64_t n4 = numel >> 2;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    s16x4_t gb = reinterpret_cast<const s16x4_t*>(g_bf16)[i];
    f32x4_t p = reinterpret_cast<const f32x4_t*>(master)[i];
    f32x4_t m = reinterpret_cast<const f32x4_t*>(m_state)[i];
    f32x4_t v = reinterpret_cast<const f32x4_t*>(v_state)[i];
    s16x4_t pb;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      float g = grad_scale * bf16_bits_to_f32(gb[c]);
      m[c] = b1 * m[c] + (1.f - b1) * g;
      v[c] = b2 * v[c] + (1.f - b2) * g * g;
      p[c] -= lr * weight_decay * p[c];  // decoupled decay
      p[c] -= lr_t * m[c] / (sqrtf(v[c]) + eps);
      pb[c] = f32_to_bf16_bits(p[c]);
    }
    reinterpret_cast<f32x4_t*>(master)[i] = p;
    reinterpret_cast<f32x4_t*>(m_state)[i] = m;
    reinterpret_cast<f32x4_t*>(v_state)[i] = v;
    reinterpret_cast<s16x4_t*>(p_bf16)[i] = pb;
  }
  if (blockIdx.x == 0) {
    for (int64_t i = (n4 << 2) + threadIdx.x; i < numel; i += blockDim.x) {
      float g = grad_scale * bf16_bits_to_f32(g_bf16[i]);
      m_state[i] = b1 * m_state[i] + (1.f - b1) * g;
      v_state[i] = b2 * v_state[i] + (1.f - b2) * g * g;
      master[i] -= lr * weight_decay * master[i];
      master[i] -= lr_t * m_state[i] / (sqrtf(v_state[i]) + eps);
      p_bf16[i] = f32_to_bf16_bits(master[i]);
    }
  }
}

extern "C" {

void edl_fused_sgd_bf16(void* p_bf16, float* master, float* vel,
                        const void* g_bf16, int64_t numel, float lr, float mu,
                        bool nesterov, float weight_decay, float grad_scale,
                        hipStream_t s) {
  fused_sgd_bf16_kernel<<<grid_for(numel >> 2), THREADS, 0, s>>>(
      reinterpret_cast<short*>(p_bf16), master, vel,
      reinterpret_cast<const short*>(g_bf16), numel, lr, mu, nesterov,
      weight_decay, grad_scale);
}

void edl_fused_adamw_bf16(void* p_bf16, float* master, float* m, float* v,
                          const void* g_bf16, int64_t numel, float lr_t,
                          float b1, float b2, float eps, float weight_decay,
                          float lr, float grad_scale, hipStream_t s) {
  fused_adamw_bf16_kernel<<<grid_for(numel >> 2), THREADS, 0, s>>>(
      reinterpret_cast<short*>(p_bf16), master, m, v,
      reinterpret_cast<const short*>(g_bf16), numel, lr_t, b1, b2, eps,
      weight_decay, lr, grad_scale);
}

}  // extern "C"
