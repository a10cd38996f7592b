// Fused AdamW step for bf16 training (gfx950).
//
// One elementwise kernel updates, per parameter tensor: the fp32 master
// copy, the fp32 exp_avg / exp_avg_sq state, AND the live bf16 parameter
// (cast of the master) in a single pass — torch.optim.AdamW on bf16 params
// runs 6+ elementwise kernels per tensor and keeps bf16 state (which loses
// the small-update tail entirely: bf16 has 8 mantissa bits).
//
// Math (decoupled weight decay, PyTorch AdamW semantics):
//   m = b1*m + (1-b1)*g ; v = b2*v + (1-b2)*g^2
//   mhat = m/(1-b1^t)   ; vhat = v/(1-b2^t)
//   master -= lr * (mhat / (sqrt(vhat) + eps) + wd * master)
//   param   = bf16(master)
// Grad may be bf16 (DDP-bucketed) or f32.

#include "common.h"

namespace {

DEV float tof(bf16 x) { return bf2f(x); }
DEV float tof(float x) { return x; }

template <typename GT>
__global__ void adamw_kernel(bf16* __restrict__ param,
                             const GT* __restrict__ grad,
                             float* __restrict__ master,
                             float* __restrict__ m, float* __restrict__ v,
                             long n, float lr, float b1, float b2, float eps,
                             float wd, float bc1, float bc2) {
  const long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = i0; i < n; i += stride) {
    const float g = tof(grad[i]);
    const float mi = b1 * m[i] + (1.f - b1) * g;
    const float vi = b2 * v[i] + (1.f - b2) * g * g;
    m[i] = mi;
    v[i] = vi;
    const float mhat = mi / bc1;
    const float vhat = vi / bc2;
    float p = master[i];
    p -= lr * (mhat / (sqrtf(vhat) + eps) + wd * p);
    master[i] = p;
    param[i] = f2bf(p);
  }
}

}  // namespace

void launch_adamw(bf16* param, const void* grad, bool grad_bf16, float* master,
                  float* m, float* v, long n, float lr, float b1, float b2,
                  float eps, float wd, long step, hipStream_t stream) {
  const int block = 256;
  long want = (n + 4 * block - 1) / (4 * block);
  const int grid = (int)(want < 2048 ? (want < 1 ? 1 : want) : 2048);
  const float bc1 = 1.f - powf(b1, (float)step);
  const float bc2 = 1.f - powf(b2, (float)step);
  if (grad_bf16)
    hipLaunchKernelGGL((adamw_kernel<bf16>), dim3(grid), dim3(block), 0,
                       stream, param, (const bf16*)grad, master, m, v, n, lr,
                       b1, b2, eps, wd, bc1, bc2);
  else
    hipLaunchKernelGGL((adamw_kernel<float>), dim3(grid), dim3(block), 0,
                       stream, param, (const float*)grad, master, m, v, n, lr,
                       b1, b2, eps, wd, bc1, bc2);
}
