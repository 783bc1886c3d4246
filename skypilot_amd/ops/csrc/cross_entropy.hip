// Fused cross-entropy (forward loss + in-place logit gradient) for CDNA4.
//
// For Llama-3's 128256-token vocab the logits tensor dominates memory
// traffic, so forward and backward are fused into one kernel that reads
// the logits twice and overwrites them with the gradient — the softmax
// matrix is never materialized separately (reference has no counterpart;
// this is the bundled-trainer hot op).
//
// logits [N, V] bf16 (overwritten with d_logits * grad_scale),
// targets [N] int32, loss [N] fp32; rows with target == ignore_index get
// zero loss/grad.
#include "common.h"

extern "C" __global__ void cross_entropy_kernel(
    unsigned short* __restrict__ logits, const int* __restrict__ targets,
    float* __restrict__ loss, long long N, int V, float grad_scale,
    int ignore_index) {
  __shared__ float red[16];
  const int nvec = V / 8;
  const int rem = V - nvec * 8;

  for (long long row = blockIdx.x; row < N; row += gridDim.x) {
    unsigned short* lrow = logits + row * V;
    const int tgt = targets[row];
    if (tgt == ignore_index) {
      for (int v = threadIdx.x; v < nvec; v += blockDim.x)
        ((s16x8*)lrow)[v] = s16x8{0, 0, 0, 0, 0, 0, 0, 0};
      for (int j = nvec * 8 + threadIdx.x; j < V; j += blockDim.x) lrow[j] = 0;
      if (threadIdx.x == 0) loss[row] = 0.f;
      __syncthreads();
      continue;
    }
    // Pass 1: row max.
    float mx = -INFINITY;
    for (int v = threadIdx.x; v < nvec; v += blockDim.x) {
      s16x8 x = ((const s16x8*)lrow)[v];
      float f[8];
      bf8_to_f32(x, f);
#pragma unroll
      for (int i = 0; i < 8; ++i) mx = fmaxf(mx, f[i]);
    }
    for (int j = nvec * 8 + threadIdx.x; j < V; j += blockDim.x)
      mx = fmaxf(mx, bf2f(lrow[j]));
    mx = block_reduce_max(mx, red);
    // Pass 2: sum of exp (logits still in L2/L3 for modest N).
    float se = 0.f;
    for (int v = threadIdx.x; v < nvec; v += blockDim.x) {
      s16x8 x = ((const s16x8*)lrow)[v];
      float f[8];
      bf8_to_f32(x, f);
#pragma unroll
      for (int i = 0; i < 8; ++i) se += __expf(f[i] - mx);
    }
    for (int j = nvec * 8 + threadIdx.x; j < V; j += blockDim.x)
      se += __expf(bf2f(lrow[j]) - mx);
    se = block_reduce_sum(se, red);
    const float logsum = __logf(se) + mx;
    const float inv_se = 1.f / se;
    // Pass 3: gradient written in place: (softmax - onehot) * grad_scale.
    float tgt_logit = 0.f;
    for (int v = threadIdx.x; v < nvec; v += blockDim.x) {
      s16x8 x = ((const s16x8*)lrow)[v];
      float f[8];
      bf8_to_f32(x, f);
      s16x8 o;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        int col = v * 8 + i;
        float p = __expf(f[i] - mx) * inv_se;
        if (col == tgt) {
          tgt_logit = f[i];
          p -= 1.f;
        }
        o[i] = (short)f2bf(p * grad_scale);
      }
      ((s16x8*)lrow)[v] = o;
    }
    for (int j = nvec * 8 + threadIdx.x; j < V; j += blockDim.x) {
      float f = bf2f(lrow[j]);
      float p = __expf(f - mx) * inv_se;
      if (j == tgt) {
        tgt_logit = f;
        p -= 1.f;
      }
      lrow[j] = f2bf(p * grad_scale);
    }
    // Exactly one thread saw the target column.
    (void)rem;
    float tl = block_reduce_sum(tgt_logit, red);
    if (threadIdx.x == 0) loss[row] = logsum - tl;
    __syncthreads();
  }
}

extern "C" void cross_entropy_launch(void* logits, const int* targets,
                                     float* loss, long long N, int V,
                                     float grad_scale, int ignore_index,
                                     hipStream_t stream) {
  int grid = N < 2048 ? (int)N : 2048;
  hipLaunchKernelGGL(cross_entropy_kernel, dim3(grid), dim3(256), 0, stream,
                     (unsigned short*)logits, targets, loss, N, V, grad_scale,
                     ignore_index);
}
