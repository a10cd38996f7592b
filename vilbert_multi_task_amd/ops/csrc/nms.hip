// Batched multi-class greedy NMS for the region-feature extractor.
//
// The reference runs a Python loop over 1600 classes, one CUDA nms() call
// each (/root/reference/worker.py:145-154) — the serving-path bottleneck
// SURVEY.md flags ("GPU hot loop #2 + Python-loop bottleneck"). Here ALL
// classes run in one launch: one workgroup per class, boxes staged in LDS,
// the greedy suppression walk parallelized across the wave lanes
// (64-wide ballot, cdna_hip_programming.md §1).
//
// Inputs follow the worker contract: one shared proposal box set [R,4]
// (x1,y1,x2,y2) and per-class scores [R,C]; `order` is the per-class
// descending score order (argsort done on-device by torch.sort — one call,
// not 1600). Output: scores with suppressed entries zeroed, so the Python
// side reproduces worker.py:156-174 (max over classes, top-K) with two
// tensor ops.

#include "common.h"

#define NMS_MAX_R 1024  // proposals per image (reference uses ~1000)

__global__ void nms_multiclass_kernel(const float* __restrict__ boxes,   // [R,4]
                                      const float* __restrict__ scores,  // [R,C]
                                      const long* __restrict__ order,    // [R,C] per-class descending
                                      float* __restrict__ out,           // [R,C] zeroed-if-suppressed
                                      int R, int C, float iou_thr,
                                      float score_thr) {
  const int c = blockIdx.x;
  if (c >= C) return;
  const int tid = threadIdx.x;

  __shared__ float4 sbox[NMS_MAX_R];
  __shared__ unsigned char suppressed[NMS_MAX_R];
  for (int i = tid; i < R; i += blockDim.x) {
    sbox[i] = make_float4(boxes[i * 4 + 0], boxes[i * 4 + 1], boxes[i * 4 + 2],
                          boxes[i * 4 + 3]);
    suppressed[i] = scores[(long)order[(long)i * C + c] * C + c] <= score_thr;
  }
  __syncthreads();

  // greedy walk in score order; inner IoU test parallel over threads
  for (int i = 0; i < R - 1; ++i) {
    __syncthreads();
    if (suppressed[i]) continue;
    const int bi = (int)order[(long)i * C + c];
    const float4 a = sbox[bi];
    const float areaA = fmaxf(a.z - a.x, 0.f) * fmaxf(a.w - a.y, 0.f);
    for (int j = i + 1 + tid; j < R; j += blockDim.x) {
      if (suppressed[j]) continue;
      const int bj = (int)order[(long)j * C + c];
      const float4 bb = sbox[bj];
      const float ix = fminf(a.z, bb.z) - fmaxf(a.x, bb.x);
      const float iy = fminf(a.w, bb.w) - fmaxf(a.y, bb.y);
      const float inter = fmaxf(ix, 0.f) * fmaxf(iy, 0.f);
      const float areaB = fmaxf(bb.z - bb.x, 0.f) * fmaxf(bb.w - bb.y, 0.f);
      const float iou = inter / (areaA + areaB - inter);
      if (iou > iou_thr) suppressed[j] = 1;
    }
  }
  __syncthreads();

  for (int i = tid; i < R; i += blockDim.x) {
    const long bi = order[(long)i * C + c];
    out[bi * C + c] = suppressed[i] ? 0.f : scores[bi * C + c];
  }
}

void launch_nms_multiclass(const float* boxes, const float* scores,
                           const long* order, float* out, int R, int C,
                           float iou_thr, float score_thr, hipStream_t stream) {
  hipLaunchKernelGGL(nms_multiclass_kernel, dim3(C), dim3(256), 0, stream, boxes,
                     scores, order, out, R, C, iou_thr, score_thr);
}
