// Fused AdamW step for gfx950.
//
// Replaces the reference's per-module torch.optim step executed on each
// worker (tensorlink/ml/worker.py:1279-1297). The Python-side optimizer
// flattens a stage's parameters into ONE contiguous buffer, so the whole
// stage steps in a single launch (cf. SURVEY.md §2.1 "one fused HIP Adam
// kernel launch per stage rank").
//
// param: bf16 or fp32 [n]; grad: same dtype as param; exp_avg/exp_avg_sq:
// fp32 [n]. Decoupled weight decay, bias correction — matches
// torch.optim.AdamW numerics (fp32 math throughout).

#include "common.hpp"

namespace {

constexpr int BLOCK = 256;

template <typename T, int VEC>
__global__ __launch_bounds__(BLOCK) void adamw_kernel(
    T* __restrict__ param, const T* __restrict__ grad,
    float* __restrict__ m, float* __restrict__ v, int64_t n, float lr,
    float beta1, float beta2, float eps, float weight_decay, float bc1,
    float bc2) {
  const int64_t base = ((int64_t)blockIdx.x * BLOCK + threadIdx.x) * VEC;
  if (base >= n) return;
#pragma unroll
  for (int j = 0; j < VEC; ++j) {
    const int64_t i = base + j;
    if (i >= n) break;
    float g;
    if constexpr (sizeof(T) == 2) g = bf2f(((const bf16*)grad)[i]);
    else g = ((const float*)grad)[i];
    float mi = m[i] * beta1 + (1.f - beta1) * g;
    float vi = v[i] * beta2 + (1.f - beta2) * g * g;
    m[i] = mi;
    v[i] = vi;
    const float denom = sqrtf(vi / bc2) + eps;
    float p;
    if constexpr (sizeof(T) == 2) p = bf2f(((const bf16*)param)[i]);
    else p = ((const float*)param)[i];
    p = p * (1.f - lr * weight_decay) - lr * (mi / bc1) / denom;
    if constexpr (sizeof(T) == 2) ((bf16*)param)[i] = f2bf(p);
    else ((float*)param)[i] = p;
  }
}

}  // namespace

extern "C" {

void tl_adamw(void* param, const void* grad, void* m, void* v, int64_t n,
              int is_bf16, float lr, float beta1, float beta2, float eps,
              float weight_decay, int step, hipStream_t stream) {
  const float bc1 = 1.f - powf(beta1, (float)step);
  const float bc2 = 1.f - powf(beta2, (float)step);
  constexpr int VEC = 4;
  const int64_t work = (n + VEC - 1) / VEC;
  dim3 grid((uint32_t)((work + BLOCK - 1) / BLOCK)), block(BLOCK);
  if (is_bf16)
    hipLaunchKernelGGL((adamw_kernel<bf16, VEC>), grid, block, 0, stream,
                       (bf16*)param, (const bf16*)grad, (float*)m, (float*)v,
                       n, lr, beta1, beta2, eps, weight_decay, bc1, bc2);
  else
    hipLaunchKernelGGL((adamw_kernel<float, VEC>), grid, block, 0, stream,
                       (float*)param, (const float*)grad, (float*)m,
                       (float*)v, n, lr, beta1, beta2, eps, weight_decay, bc1,
                       bc2);
}

}  // extern "C"
