// Fused SwiGLU elementwise kernels: y = silu(g) * u (Llama MLP gate).
// Eager costs silu (read g, write s) + mul (read s,u, write y) forward and a
// longer chain backward; fused is one streaming pass each way (bf16 in/out,
// fp32 math, 16 B/lane).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define SG_BLOCK 256

namespace {

typedef __hip_bfloat16 bf16;

union sg_bf16x8 {
  uint4 u4;
  unsigned short h[8];
};

__device__ __forceinline__ float sg_b2f(unsigned short v) {
  __hip_bfloat16_raw r;
  r.x = v;
  return __bfloat162float(*reinterpret_cast<bf16*>(&r));
}

__device__ __forceinline__ unsigned short sg_f2b(float f) {
  bf16 h = __float2bfloat16(f);
  return *reinterpret_cast<unsigned short*>(&h);
}

__global__ __launch_bounds__(SG_BLOCK) void swiglu_fwd_kernel(
    const sg_bf16x8* __restrict__ g, const sg_bf16x8* __restrict__ u,
    sg_bf16x8* __restrict__ y, long n8) {
  long stride = (long)gridDim.x * SG_BLOCK;
  for (long i = (long)blockIdx.x * SG_BLOCK + threadIdx.x; i < n8;
       i += stride) {
    sg_bf16x8 vg = g[i], vu = u[i], o;
    #pragma unroll
    for (int k = 0; k < 8; ++k) {
      float x = sg_b2f(vg.h[k]);
      float sig = 1.f / (1.f + __expf(-x));
      o.h[k] = sg_f2b(x * sig * sg_b2f(vu.h[k]));
    }
    y[i] = o;
  }
}

__global__ __launch_bounds__(SG_BLOCK) void swiglu_bwd_kernel(
    const sg_bf16x8* __restrict__ dy, const sg_bf16x8* __restrict__ g,
    const sg_bf16x8* __restrict__ u, sg_bf16x8* __restrict__ dg,
    sg_bf16x8* __restrict__ du, long n8) {
  long stride = (long)gridDim.x * SG_BLOCK;
  for (long i = (long)blockIdx.x * SG_BLOCK + threadIdx.x; i < n8;
       i += stride) {
    sg_bf16x8 vdy = dy[i], vg = g[i], vu = u[i], og, ou;
    #pragma unroll
    for (int k = 0; k < 8; ++k) {
      float x = sg_b2f(vg.h[k]);
      float d = sg_b2f(vdy.h[k]);
      float uu = sg_b2f(vu.h[k]);
      float sig = 1.f / (1.f + __expf(-x));
      float si = x * sig;
      og.h[k] = sg_f2b(d * uu * (sig * (1.f + x * (1.f - sig))));
      ou.h[k] = sg_f2b(d * si);
    }
    dg[i] = og;
    du[i] = ou;
  }
}

}  // namespace

at::Tensor swiglu_fwd(at::Tensor g, at::Tensor u) {
  TORCH_CHECK(g.scalar_type() == at::kBFloat16 && g.is_contiguous() &&
                  u.is_contiguous() && g.numel() == u.numel() &&
                  g.numel() % 8 == 0,
              "swiglu: contiguous bf16, numel %% 8 == 0");
  auto y = at::empty_like(g);
  const long n8 = g.numel() / 8;
  auto stream = at::hip::getCurrentHIPStream().stream();
  int grid = (int)std::min<long>((n8 + SG_BLOCK - 1) / SG_BLOCK, 4096);
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(grid), dim3(SG_BLOCK), 0, stream,
                     (const sg_bf16x8*)g.data_ptr(),
                     (const sg_bf16x8*)u.data_ptr(), (sg_bf16x8*)y.data_ptr(),
                     n8);
  return y;
}

std::vector<at::Tensor> swiglu_bwd(at::Tensor dy, at::Tensor g, at::Tensor u) {
  auto dg = at::empty_like(g);
  auto du = at::empty_like(u);
  const long n8 = g.numel() / 8;
  auto stream = at::hip::getCurrentHIPStream().stream();
  int grid = (int)std::min<long>((n8 + SG_BLOCK - 1) / SG_BLOCK, 4096);
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(grid), dim3(SG_BLOCK), 0, stream,
                     (const sg_bf16x8*)dy.data_ptr(),
                     (const sg_bf16x8*)g.data_ptr(),
                     (const sg_bf16x8*)u.data_ptr(),
                     (sg_bf16x8*)dg.data_ptr(), (sg_bf16x8*)du.data_ptr(),
                     n8);
  return {dg, du};
}
