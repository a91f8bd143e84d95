// Fused RMSNorm (Llama-style) training kernels for MI355X.
//
// Eager RMSNorm (benchmarks/models.py RMSNorm / reference-free) costs ~5
// kernels per call with an fp32 up-cast of the whole activation (x.float()),
// twice per decoder layer, forward and backward.  These kernels do it in one
// streaming pass each way over the [T, D] token matrix (bf16, fp32
// accumulation, 16 B/lane vectorized):
//   forward : per-row sumsq -> r = rsqrt(mean+eps) (saved, fp32 [T]),
//             y = x * r * w
//   backward: per-row s = sum(dy*w*x);  dx = r*(dy*w) - x * r^3/D * s;
//             dw accumulated per-block in LDS -> [nblocks, D] partials ->
//             second-stage reduce (same no-global-atomics pattern as the
//             fused BN kernels, csrc/fused_bn.hip).
//
// Layout contract: x is [T, D] row-major bf16 (a view of [B, S, D]),
// D % 8 == 0, D <= 16384 (LDS partial row = D floats <= 64 KB).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define RN_BLOCK 256

namespace {

typedef __hip_bfloat16 bf16;

union rn_bf16x8 {
  uint4 u4;
  unsigned short h[8];
};

__device__ __forceinline__ float rn_b2f(unsigned short v) {
  __hip_bfloat16_raw r;
  r.x = v;
  return __bfloat162float(*reinterpret_cast<bf16*>(&r));
}

__device__ __forceinline__ unsigned short rn_f2b(float f) {
  bf16 h = __float2bfloat16(f);
  return *reinterpret_cast<unsigned short*>(&h);
}

__device__ __forceinline__ float rn_block_reduce(float v) {
  __shared__ float smem[RN_BLOCK / 64];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off);
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  v = (threadIdx.x < RN_BLOCK / 64) ? smem[threadIdx.x] : 0.f;
  if (wid == 0) {
    #pragma unroll
    for (int off = RN_BLOCK / 128; off > 0; off >>= 1)
      v += __shfl_down(v, off);
  }
  return v;  // valid in thread 0
}

// One block per token row.
__global__ __launch_bounds__(RN_BLOCK) void rms_fwd_kernel(
    const rn_bf16x8* __restrict__ x, const rn_bf16x8* __restrict__ w,
    rn_bf16x8* __restrict__ y, float* __restrict__ invr, int c8, int D,
    float eps) {
  const long base = (long)blockIdx.x * c8;
  float ss = 0.f;
  for (int slot = threadIdx.x; slot < c8; slot += RN_BLOCK) {
    rn_bf16x8 v = x[base + slot];
    #pragma unroll
    for (int k = 0; k < 8; ++k) {
      float f = rn_b2f(v.h[k]);
      ss += f * f;
    }
  }
  ss = rn_block_reduce(ss);
  __shared__ float s_r;
  if (threadIdx.x == 0) {
    float r = rsqrtf(ss / (float)D + eps);
    s_r = r;
    invr[blockIdx.x] = r;
  }
  __syncthreads();
  const float r = s_r;
  for (int slot = threadIdx.x; slot < c8; slot += RN_BLOCK) {
    rn_bf16x8 v = x[base + slot];  // L2 hit: just read in pass 1
    rn_bf16x8 vw = w[slot];
    rn_bf16x8 o;
    #pragma unroll
    for (int k = 0; k < 8; ++k)
      o.h[k] = rn_f2b(rn_b2f(v.h[k]) * r * rn_b2f(vw.h[k]));
    y[base + slot] = o;
  }
}

// Blocks own contiguous row ranges; dw accumulates in LDS (one slot per
// thread stride, no conflicts), flushed once to the per-block partial row.
__global__ __launch_bounds__(RN_BLOCK) void rms_bwd_kernel(
    const rn_bf16x8* __restrict__ x, const rn_bf16x8* __restrict__ dy,
    const rn_bf16x8* __restrict__ w, const float* __restrict__ invr,
    rn_bf16x8* __restrict__ dx, float* __restrict__ dw_part, long T, int c8,
    int D, int rows_per_block) {
  extern __shared__ float s_dw[];  // c8*8 floats
  for (int i = threadIdx.x; i < c8 * 8; i += RN_BLOCK) s_dw[i] = 0.f;
  __syncthreads();
  long row0 = (long)blockIdx.x * rows_per_block;
  long row_end = min(row0 + rows_per_block, T);
  for (long row = row0; row < row_end; ++row) {
    const long base = row * c8;
    const float r = invr[row];
    float s = 0.f;
    for (int slot = threadIdx.x; slot < c8; slot += RN_BLOCK) {
      rn_bf16x8 vdy = dy[base + slot];
      rn_bf16x8 vx = x[base + slot];
      rn_bf16x8 vw = w[slot];
      #pragma unroll
      for (int k = 0; k < 8; ++k)
        s += rn_b2f(vdy.h[k]) * rn_b2f(vw.h[k]) * rn_b2f(vx.h[k]);
    }
    s = rn_block_reduce(s);
    __shared__ float s_coef;
    if (threadIdx.x == 0) s_coef = r * r * r / (float)D * s;
    __syncthreads();
    const float coef = s_coef;
    for (int slot = threadIdx.x; slot < c8; slot += RN_BLOCK) {
      rn_bf16x8 vdy = dy[base + slot];
      rn_bf16x8 vx = x[base + slot];
      rn_bf16x8 vw = w[slot];
      rn_bf16x8 o;
      #pragma unroll
      for (int k = 0; k < 8; ++k) {
        float gdy = rn_b2f(vdy.h[k]);
        float fx = rn_b2f(vx.h[k]);
        o.h[k] = rn_f2b(r * gdy * rn_b2f(vw.h[k]) - fx * coef);
        s_dw[slot * 8 + k] += gdy * fx * r;
      }
      dx[base + slot] = o;
    }
    __syncthreads();
  }
  float* out = dw_part + (long)blockIdx.x * D;
  for (int i = threadIdx.x; i < c8 * 8; i += RN_BLOCK) out[i] = s_dw[i];
}

// Reduce [nblocks, D] partials -> dw (bf16, matching the weight dtype).
#define RN_PPAR 16
#define RN_CH 64
__global__ __launch_bounds__(RN_PPAR * RN_CH) void rms_dw_reduce_kernel(
    const float* __restrict__ part, int nblocks, bf16* __restrict__ dw,
    int D) {
  const int c = blockIdx.x * RN_CH + threadIdx.x % RN_CH;
  const int pp = threadIdx.x / RN_CH;
  float s = 0.f;
  if (c < D) {
    for (int p = pp; p < nblocks; p += RN_PPAR)
      s += part[(long)p * D + c];
  }
  __shared__ float smem[RN_PPAR * RN_CH];
  smem[pp * RN_CH + threadIdx.x % RN_CH] = s;
  __syncthreads();
  if (pp == 0 && c < D) {
    #pragma unroll
    for (int r = 1; r < RN_PPAR; ++r) s += smem[r * RN_CH + threadIdx.x % RN_CH];
    dw[c] = __float2bfloat16(s);
  }
}

}  // namespace

std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor w, double eps) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
                  w.scalar_type() == at::kBFloat16,
              "rmsnorm: bf16 only");
  const long D = x.size(-1);
  TORCH_CHECK(D % 8 == 0 && D <= 16384, "rmsnorm: D must be %8==0, <=16384");
  const long T = x.numel() / D;
  const int c8 = (int)(D / 8);
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto y = at::empty_like(x);
  auto invr = at::empty({T}, x.options().dtype(at::kFloat));
  hipLaunchKernelGGL(rms_fwd_kernel, dim3((unsigned)T), dim3(RN_BLOCK), 0,
                     stream, (const rn_bf16x8*)x.data_ptr(),
                     (const rn_bf16x8*)w.data_ptr(), (rn_bf16x8*)y.data_ptr(),
                     invr.data_ptr<float>(), c8, (int)D, (float)eps);
  return {y, invr};
}

std::vector<at::Tensor> rmsnorm_bwd(at::Tensor x, at::Tensor dy, at::Tensor w,
                                    at::Tensor invr) {
  const long D = x.size(-1);
  const long T = x.numel() / D;
  const int c8 = (int)(D / 8);
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto dx = at::empty_like(x);
  // 1024 blocks x 4 waves = 16 waves/CU; LDS dw row = D*4 B <= 64 KB
  long target = 1024;
  int rpb = (int)((T + target - 1) / target);
  if (rpb < 1) rpb = 1;
  int nblocks = (int)((T + rpb - 1) / rpb);
  auto part = at::empty({(long)nblocks * D}, x.options().dtype(at::kFloat));
  size_t lds = (size_t)D * sizeof(float);
  hipLaunchKernelGGL(rms_bwd_kernel, dim3(nblocks), dim3(RN_BLOCK), lds,
                     stream, (const rn_bf16x8*)x.data_ptr(),
                     (const rn_bf16x8*)dy.data_ptr(),
                     (const rn_bf16x8*)w.data_ptr(), invr.data_ptr<float>(),
                     (rn_bf16x8*)dx.data_ptr(), part.data_ptr<float>(), T,
                     c8, (int)D, rpb);
  auto dw = at::empty_like(w);
  hipLaunchKernelGGL(rms_dw_reduce_kernel,
                     dim3((unsigned)((D + RN_CH - 1) / RN_CH)),
                     dim3(RN_PPAR * RN_CH), 0, stream,
                     part.data_ptr<float>(), nblocks,
                     (bf16*)dw.data_ptr(), (int)D);
  return {dx, dw};
}
