#include "hip/hip_runtime.h"
// Fused AdamW over flat parameter buffers, CDNA4.
//
// The training loop keeps ONE flat fp32 master copy + ONE flat bf16 working
// copy of all parameters (models/llama.py builds views into them), so the
// optimizer is a single fused elementwise pass:
//   g   = bf16 grad * grad_scale            (grad_scale folds in the 1/world
//                                            averaging of the all-reduce)
//   m   = b1*m + (1-b1)*g ;  v = b2*v + (1-b2)*g^2
//   p  -= lr * ( (m*bc1) / (sqrt(v*bc2)+eps) + wd*p )
//   p_bf16 = bf16(p)
// Traffic: 30 B/element, one pass — HBM-bound by design.
#include "common.hip.h"

extern "C" __global__ void __launch_bounds__(256)
adamw_kernel(float* __restrict__ p, ushort_t* __restrict__ p_bf16,
             const ushort_t* __restrict__ g, float* __restrict__ m,
             float* __restrict__ v, int64_t n4, float lr, float b1, float b2,
             float eps, float wd, float bc1, float bc2, float grad_scale) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += (int64_t)gridDim.x * blockDim.x) {
    floatx4 pv = reinterpret_cast<floatx4*>(p)[i];
    floatx4 mv = reinterpret_cast<floatx4*>(m)[i];
    floatx4 vv = reinterpret_cast<floatx4*>(v)[i];
    ushortx4 gb = reinterpret_cast<const ushortx4*>(g)[i];
    ushortx4 ob;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gf = bf2f(gb[j]) * grad_scale;
      mv[j] = b1 * mv[j] + (1.f - b1) * gf;
      vv[j] = b2 * vv[j] + (1.f - b2) * gf * gf;
      float update = (mv[j] * bc1) / (sqrtf(vv[j] * bc2) + eps) + wd * pv[j];
      pv[j] -= lr * update;
      ob[j] = f2bf(pv[j]);
    }
    reinterpret_cast<floatx4*>(p)[i] = pv;
    reinterpret_cast<floatx4*>(m)[i] = mv;
    reinterpret_cast<floatx4*>(v)[i] = vv;
    reinterpret_cast<ushortx4*>(p_bf16)[i] = ob;
  }
}

extern "C" void launch_adamw(float* p, void* p_bf16, const void* g, float* m,
                             float* v, int64_t n, float lr, float b1, float b2,
                             float eps, float wd, float bc1, float bc2,
                             float grad_scale, hipStream_t s) {
  int64_t n4 = n / 4;
  int64_t blocks = (n4 + 255) / 256;
  if (blocks > 16384) blocks = 16384;
  hipLaunchKernelGGL(adamw_kernel, dim3((uint32_t)blocks), dim3(256), 0, s, p,
                     (ushort_t*)p_bf16, (const ushort_t*)g, m, v, n4, lr, b1,
                     b2, eps, wd, bc1, bc2, grad_scale);
}
