// Fused fp8 (OCP e4m3/e5m2) quantization kernels — ROUND-2 WORK IN
// PROGRESS (see NOTES.md "fp8 end-to-end is slower than bf16").
//
// Status: compiles for gfx950; exercised only by env-gated tests
// (STOKE_FP8V2_TEST=1 in tests/test_fa_wip.py-style gating).  The active
// fp8 path (stoke/nn/fp8.py) still uses the validated per-call amax
// version; round 2 swaps it for delayed scaling built on these kernels.
//
// What the v1 profile showed: per-GEMM overhead = one full amax reduction
// pass + a separate quantize pass per operand + `.t().contiguous()` copies
// in backward.  These kernels collapse all of it:
//   fp8_quant    : one pass  bf16 -> fp8 with a GIVEN scale, while
//                  accumulating the NEXT amax as a byproduct (delayed
//                  scaling needs no extra reduction pass).
//   fp8_quant_t  : same, but emits BOTH row-major and transposed fp8
//                  copies (LDS-tiled 32x32 transpose) so backward's
//                  column-major operands need no `.t().contiguous()`.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define FQ_BLOCK 256

namespace {

typedef __hip_bfloat16 bf16;

union fq_bf16x8 {
  uint4 u4;
  unsigned short h[8];
};

__device__ __forceinline__ float fq_b2f(unsigned short v) {
  __hip_bfloat16_raw r;
  r.x = v;
  return __bfloat162float(*reinterpret_cast<bf16*>(&r));
}

template <typename F8>
__device__ __forceinline__ unsigned char fq_cast(float f) {
  F8 q(f);
  return *reinterpret_cast<unsigned char*>(&q);
}

// atomic max for non-negative floats via ordered uint bits
__device__ __forceinline__ void fq_atomic_amax(float* addr, float v) {
  atomicMax(reinterpret_cast<unsigned int*>(addr),
            __float_as_uint(v));
}

// y8[i] = fp8(x[i] / *scale); *amax_next = max|x| (block-reduced)
template <typename F8>
__global__ __launch_bounds__(FQ_BLOCK) void fp8_quant_kernel(
    const fq_bf16x8* __restrict__ x, unsigned char* __restrict__ y,
    const float* __restrict__ scale, float* __restrict__ amax_next,
    long n8, float fp8_max) {
  const float inv = 1.f / *scale;
  float am = 0.f;
  long stride = (long)gridDim.x * FQ_BLOCK;
  for (long i = (long)blockIdx.x * FQ_BLOCK + threadIdx.x; i < n8;
       i += stride) {
    fq_bf16x8 vv = x[i];
    unsigned char o[8];
    #pragma unroll
    for (int e = 0; e < 8; ++e) {
      float f = fq_b2f(vv.h[e]);
      am = fmaxf(am, fabsf(f));
      o[e] = fq_cast<F8>(fminf(fmaxf(f * inv, -fp8_max), fp8_max));
    }
    *reinterpret_cast<uint2*>(y + i * 8) = *reinterpret_cast<uint2*>(o);
  }
  // block-reduce the amax, one atomic per block
  __shared__ float smem[FQ_BLOCK / 64];
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    am = fmaxf(am, __shfl_down(am, off));
  if ((threadIdx.x & 63) == 0) smem[threadIdx.x >> 6] = am;
  __syncthreads();
  if (threadIdx.x == 0) {
    #pragma unroll
    for (int w = 1; w < FQ_BLOCK / 64; ++w) am = fmaxf(am, smem[w]);
    fq_atomic_amax(amax_next, am);
  }
}

// Row-major [M,N] -> y8 row-major AND yt8 [N,M]; 32x32 LDS transpose tiles.
// Block = 256 threads = 8 rows of 32 lanes per pass; grid 2-D over tiles.
template <typename F8>
__global__ __launch_bounds__(FQ_BLOCK) void fp8_quant_t_kernel(
    const bf16* __restrict__ x, unsigned char* __restrict__ y,
    unsigned char* __restrict__ yt, const float* __restrict__ scale,
    float* __restrict__ amax_next, int M, int N, float fp8_max) {
  __shared__ unsigned char tile[32][33];  // +1 pad: bank-conflict-free
  const float inv = 1.f / *scale;
  const int tx = threadIdx.x & 31;   // col within tile
  const int ty = threadIdx.x >> 5;   // row group (8 rows/pass, 4 passes)
  const int col0 = blockIdx.x * 32;
  const int row0 = blockIdx.y * 32;
  float am = 0.f;
  #pragma unroll
  for (int p = 0; p < 4; ++p) {
    const int r = row0 + p * 8 + ty;
    const int c = col0 + tx;
    unsigned char q = 0;
    if (r < M && c < N) {
      float f = fq_b2f(*reinterpret_cast<const unsigned short*>(
          &x[(long)r * N + c]));
      am = fmaxf(am, fabsf(f));
      q = fq_cast<F8>(fminf(fmaxf(f * inv, -fp8_max), fp8_max));
      y[(long)r * N + c] = q;
    }
    tile[p * 8 + ty][tx] = q;
  }
  __syncthreads();
  #pragma unroll
  for (int p = 0; p < 4; ++p) {
    const int tr = p * 8 + ty;           // row within the TRANSPOSED tile
    const int c = col0 + tr;             // original col -> yt row
    const int r = row0 + tx;             // original row -> yt col
    if (c < N && r < M)
      yt[(long)c * M + r] = tile[tx][tr];
  }
  // amax reduction (same pattern as fp8_quant_kernel)
  __shared__ float smem[FQ_BLOCK / 64];
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    am = fmaxf(am, __shfl_down(am, off));
  if ((threadIdx.x & 63) == 0) smem[threadIdx.x >> 6] = am;
  __syncthreads();
  if (threadIdx.x == 0) {
    #pragma unroll
    for (int w = 1; w < FQ_BLOCK / 64; ++w) am = fmaxf(am, smem[w]);
    fq_atomic_amax(amax_next, am);
  }
}

}  // namespace

// dtype flag: 0 = e4m3 (max 448), 1 = e5m2 (max 57344)
at::Tensor fp8_quant(at::Tensor x, at::Tensor scale, at::Tensor amax_next,
                     int64_t kind) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.is_contiguous() &&
                  x.numel() % 8 == 0,
              "fp8_quant: contiguous bf16, numel %% 8 == 0");
  TORCH_CHECK(scale.is_cuda() && amax_next.is_cuda(),
              "fp8_quant: scale/amax must be CUDA tensors");
  auto dt = kind == 0 ? at::ScalarType::Float8_e4m3fn
                      : at::ScalarType::Float8_e5m2;
  auto y = at::empty_like(x, x.options().dtype(dt));
  const long n8 = x.numel() / 8;
  auto stream = at::hip::getCurrentHIPStream().stream();
  int grid = (int)std::min<long>((n8 + FQ_BLOCK - 1) / FQ_BLOCK, 4096);
  const float mx = kind == 0 ? 448.f : 57344.f;
  if (kind == 0)
    hipLaunchKernelGGL((fp8_quant_kernel<__hip_fp8_e4m3>), dim3(grid),
                       dim3(FQ_BLOCK), 0, stream,
                       (const fq_bf16x8*)x.data_ptr(),
                       (unsigned char*)y.data_ptr(),
                       scale.data_ptr<float>(), amax_next.data_ptr<float>(),
                       n8, mx);
  else
    hipLaunchKernelGGL((fp8_quant_kernel<__hip_fp8_e5m2>), dim3(grid),
                       dim3(FQ_BLOCK), 0, stream,
                       (const fq_bf16x8*)x.data_ptr(),
                       (unsigned char*)y.data_ptr(),
                       scale.data_ptr<float>(), amax_next.data_ptr<float>(),
                       n8, mx);
  return y;
}

std::vector<at::Tensor> fp8_quant_t(at::Tensor x, at::Tensor scale,
                                    at::Tensor amax_next, int64_t kind) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.dim() == 2 &&
                  x.is_contiguous(),
              "fp8_quant_t: contiguous 2-D bf16");
  TORCH_CHECK(scale.is_cuda() && amax_next.is_cuda() &&
                  scale.scalar_type() == at::kFloat &&
                  amax_next.scalar_type() == at::kFloat,
              "fp8_quant_t: scale/amax must be fp32 CUDA tensors");
  const int M = (int)x.size(0), N = (int)x.size(1);
  auto dt = kind == 0 ? at::ScalarType::Float8_e4m3fn
                      : at::ScalarType::Float8_e5m2;
  auto y = at::empty({M, N}, x.options().dtype(dt));
  auto yt = at::empty({N, M}, x.options().dtype(dt));
  auto stream = at::hip::getCurrentHIPStream().stream();
  dim3 grid((N + 31) / 32, (M + 31) / 32);
  const float mx = kind == 0 ? 448.f : 57344.f;
  if (kind == 0)
    hipLaunchKernelGGL((fp8_quant_t_kernel<__hip_fp8_e4m3>), grid,
                       dim3(FQ_BLOCK), 0, stream, (const bf16*)x.data_ptr(),
                       (unsigned char*)y.data_ptr(),
                       (unsigned char*)yt.data_ptr(),
                       scale.data_ptr<float>(), amax_next.data_ptr<float>(),
                       M, N, mx);
  else
    hipLaunchKernelGGL((fp8_quant_t_kernel<__hip_fp8_e5m2>), grid,
                       dim3(FQ_BLOCK), 0, stream, (const bf16*)x.data_ptr(),
                       (unsigned char*)y.data_ptr(),
                       (unsigned char*)yt.data_ptr(),
                       scale.data_ptr<float>(), amax_next.data_ptr<float>(),
                       M, N, mx);
  return {y, yt};
}
