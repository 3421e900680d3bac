#include "hip/hip_runtime.h"
// Fused cross-entropy forward+backward over a logits chunk, bf16, CDNA4.
//
// For vocab V=128k the logits tensor dominates activation memory
// (B*S*V*2 bytes). This kernel computes, per row, in ONE launch:
//   m = max(x);  s = sum(exp(x-m));  loss = log(s) + m - x[target]
//   dlogits = (exp(x-m)/s - onehot[target]) * grad_scale     (in-place!)
// The in-place dlogits write means the fp32 softmax never materializes and
// the backward needs no second kernel. Callers chunk rows so the Python side
// (ops/cross_entropy.py) can fuse this with chunked vocab GEMMs and keep peak
// memory at chunk_size*V instead of N*V.
//
// ignore_index rows produce loss=0 and zero gradient.
#include "common.hip.h"

extern "C" __global__ void __launch_bounds__(512)
cross_entropy_fwd_bwd_kernel(ushort_t* __restrict__ logits,  // [N, V] in/out
                             const int32_t* __restrict__ targets,  // [N]
                             float* __restrict__ loss,             // [N]
                             int64_t N, int64_t V, float grad_scale,
                             int32_t ignore_index, int write_dlogits) {
  __shared__ float scratch[16];
  __shared__ float s_target_logit;
  for (int64_t row = blockIdx.x; row < N; row += gridDim.x) {
    ushort_t* x = logits + row * V;
    int32_t tgt = targets[row];
    bool ignored = (tgt == ignore_index);
    if (threadIdx.x == 0) s_target_logit = 0.f;
    __syncthreads();

    // pass 1: online max + sumexp per thread, then block combine
    float m = -INFINITY, s = 0.f;
    for (int64_t h = (int64_t)threadIdx.x * 8; h < V; h += (int64_t)blockDim.x * 8) {
      floatx8 f = bf8_to_f32x8(*reinterpret_cast<const ushortx8*>(x + h));
      if (!ignored && tgt >= h && tgt < h + 8) s_target_logit = f[tgt - h];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        float v = f[i];
        if (v > m) {
          s *= __expf(m - v);
          m = v;
        }
        s += __expf(v - m);
      }
    }
    float bm = block_reduce_max(m, scratch);
    __syncthreads();
    float bs = block_reduce_sum(s * __expf(m - bm), scratch);
    __syncthreads();

    if (threadIdx.x == 0 && loss) {
      loss[row] = ignored ? 0.f : (__logf(bs) + bm - s_target_logit);
    }

    // pass 2: dlogits in place
    if (write_dlogits) {
      float inv_s = 1.f / bs;
      float gs = ignored ? 0.f : grad_scale;
      for (int64_t h = (int64_t)threadIdx.x * 8; h < V; h += (int64_t)blockDim.x * 8) {
        floatx8 f = bf8_to_f32x8(*reinterpret_cast<const ushortx8*>(x + h));
        floatx8 o;
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          float p = __expf(f[i] - bm) * inv_s;
          float grad = p - ((!ignored && (h + i) == (int64_t)tgt) ? 1.f : 0.f);
          o[i] = grad * gs;
        }
        *reinterpret_cast<ushortx8*>(x + h) = f32x8_to_bf8(o);
      }
    }
    __syncthreads();
  }
}

extern "C" void launch_cross_entropy(void* logits, const int32_t* targets,
                                     float* loss, int64_t N, int64_t V,
                                     float grad_scale, int32_t ignore_index,
                                     int write_dlogits, hipStream_t s) {
  uint32_t grid = (uint32_t)(N < 1024 ? N : 1024);
  hipLaunchKernelGGL(cross_entropy_fwd_bwd_kernel, dim3(grid), dim3(512), 0, s,
                     (ushort_t*)logits, targets, loss, N, V, grad_scale,
                     ignore_index, write_dlogits);
}
