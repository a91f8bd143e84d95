// Fused LayerNorm training kernels for MI355X (GPT-2-style, last-dim norm).
// Same single-pass streaming structure as the fused RMSNorm
// (csrc/fused_rmsnorm.hip): bf16 in/out, fp32 accumulation, per-row mean +
// invstd saved forward; backward emits dx in one pass and accumulates
// dgamma/dbeta per block in LDS -> [nblocks, 2D] partials -> second-stage
// reduce (no global atomics).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define LN_BLOCK 256

namespace {

typedef __hip_bfloat16 bf16;

union ln_bf16x8 {
  uint4 u4;
  unsigned short h[8];
};

__device__ __forceinline__ float ln_b2f(unsigned short v) {
  __hip_bfloat16_raw r;
  r.x = v;
  return __bfloat162float(*reinterpret_cast<bf16*>(&r));
}

__device__ __forceinline__ unsigned short ln_f2b(float f) {
  bf16 h = __float2bfloat16(f);
  return *reinterpret_cast<unsigned short*>(&h);
}

// Two values (sum, sumsq) reduced at once.
__device__ __forceinline__ void ln_block_reduce2(float& a, float& b) {
  __shared__ float smem[2 * LN_BLOCK / 64];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    a += __shfl_down(a, off);
    b += __shfl_down(b, off);
  }
  if (lane == 0) {
    smem[wid] = a;
    smem[LN_BLOCK / 64 + wid] = b;
  }
  __syncthreads();
  if (wid == 0) {
    a = (threadIdx.x < LN_BLOCK / 64) ? smem[threadIdx.x] : 0.f;
    b = (threadIdx.x < LN_BLOCK / 64) ? smem[LN_BLOCK / 64 + threadIdx.x] : 0.f;
    #pragma unroll
    for (int off = LN_BLOCK / 128; off > 0; off >>= 1) {
      a += __shfl_down(a, off);
      b += __shfl_down(b, off);
    }
  }
}

// One block per token row.
__global__ __launch_bounds__(LN_BLOCK) void ln_fwd_kernel(
    const ln_bf16x8* __restrict__ x, const ln_bf16x8* __restrict__ w,
    const ln_bf16x8* __restrict__ b, ln_bf16x8* __restrict__ y,
    float* __restrict__ mean_out, float* __restrict__ invstd_out, int c8,
    int D, float eps) {
  const long base = (long)blockIdx.x * c8;
  float s = 0.f, s2 = 0.f;
  for (int slot = threadIdx.x; slot < c8; slot += LN_BLOCK) {
    ln_bf16x8 v = x[base + slot];
    #pragma unroll
    for (int k = 0; k < 8; ++k) {
      float f = ln_b2f(v.h[k]);
      s += f;
      s2 += f * f;
    }
  }
  ln_block_reduce2(s, s2);
  __shared__ float s_m, s_is;
  if (threadIdx.x == 0) {
    float m = s / (float)D;
    float var = fmaxf(s2 / (float)D - m * m, 0.f);
    float is = rsqrtf(var + eps);
    s_m = m;
    s_is = is;
    mean_out[blockIdx.x] = m;
    invstd_out[blockIdx.x] = is;
  }
  __syncthreads();
  const float m = s_m, is = s_is;
  for (int slot = threadIdx.x; slot < c8; slot += LN_BLOCK) {
    ln_bf16x8 v = x[base + slot];
    ln_bf16x8 vw = w[slot];
    ln_bf16x8 vb = b[slot];
    ln_bf16x8 o;
    #pragma unroll
    for (int k = 0; k < 8; ++k)
      o.h[k] = ln_f2b((ln_b2f(v.h[k]) - m) * is * ln_b2f(vw.h[k]) +
                      ln_b2f(vb.h[k]));
    y[base + slot] = o;
  }
}

// Blocks own contiguous row ranges; dgamma/dbeta accumulate in LDS.
__global__ __launch_bounds__(LN_BLOCK) void ln_bwd_kernel(
    const ln_bf16x8* __restrict__ x, const ln_bf16x8* __restrict__ dy,
    const ln_bf16x8* __restrict__ w, const float* __restrict__ mean,
    const float* __restrict__ invstd, ln_bf16x8* __restrict__ dx,
    float* __restrict__ part, long T, int c8, int D, int rows_per_block) {
  extern __shared__ float lds[];  // [2 * c8 * 8]: dgamma then dbeta
  float* s_dg = lds;
  float* s_db = lds + c8 * 8;
  for (int i = threadIdx.x; i < c8 * 8; i += LN_BLOCK) {
    s_dg[i] = 0.f;
    s_db[i] = 0.f;
  }
  __syncthreads();
  long row0 = (long)blockIdx.x * rows_per_block;
  long row_end = min(row0 + rows_per_block, T);
  for (long row = row0; row < row_end; ++row) {
    const long base = row * c8;
    const float m = mean[row];
    const float is = invstd[row];
    float sg = 0.f, sgx = 0.f;  // sum(g), sum(g * xhat) with g = dy*w
    for (int slot = threadIdx.x; slot < c8; slot += LN_BLOCK) {
      ln_bf16x8 vdy = dy[base + slot];
      ln_bf16x8 vx = x[base + slot];
      ln_bf16x8 vw = w[slot];
      #pragma unroll
      for (int k = 0; k < 8; ++k) {
        float g = ln_b2f(vdy.h[k]) * ln_b2f(vw.h[k]);
        float xh = (ln_b2f(vx.h[k]) - m) * is;
        sg += g;
        sgx += g * xh;
      }
    }
    ln_block_reduce2(sg, sgx);
    __shared__ float s_c1, s_c2;
    if (threadIdx.x == 0) {
      s_c1 = sg / (float)D;
      s_c2 = sgx / (float)D;
    }
    __syncthreads();
    const float c1 = s_c1, c2 = s_c2;
    for (int slot = threadIdx.x; slot < c8; slot += LN_BLOCK) {
      ln_bf16x8 vdy = dy[base + slot];
      ln_bf16x8 vx = x[base + slot];
      ln_bf16x8 vw = w[slot];
      ln_bf16x8 o;
      #pragma unroll
      for (int k = 0; k < 8; ++k) {
        float d = ln_b2f(vdy.h[k]);
        float g = d * ln_b2f(vw.h[k]);
        float xh = (ln_b2f(vx.h[k]) - m) * is;
        o.h[k] = ln_f2b(is * (g - c1 - xh * c2));
        s_dg[slot * 8 + k] += d * xh;
        s_db[slot * 8 + k] += d;
      }
      dx[base + slot] = o;
    }
    __syncthreads();
  }
  float* out = part + (long)blockIdx.x * 2 * D;
  for (int i = threadIdx.x; i < c8 * 8; i += LN_BLOCK) {
    out[i] = s_dg[i];
    out[D + i] = s_db[i];
  }
}

#define LN_PPAR 16
#define LN_CH 64
__global__ __launch_bounds__(LN_PPAR * LN_CH) void ln_dwdb_reduce_kernel(
    const float* __restrict__ part, int nblocks, bf16* __restrict__ dw,
    bf16* __restrict__ db, int D) {
  const int c = blockIdx.x * LN_CH + threadIdx.x % LN_CH;
  const int pp = threadIdx.x / LN_CH;
  float sg = 0.f, sb = 0.f;
  if (c < D) {
    for (int p = pp; p < nblocks; p += LN_PPAR) {
      sg += part[(long)p * 2 * D + c];
      sb += part[(long)p * 2 * D + D + c];
    }
  }
  __shared__ float smem[2 * LN_PPAR * LN_CH];
  smem[pp * LN_CH + threadIdx.x % LN_CH] = sg;
  smem[LN_PPAR * LN_CH + pp * LN_CH + threadIdx.x % LN_CH] = sb;
  __syncthreads();
  if (pp == 0 && c < D) {
    #pragma unroll
    for (int r = 1; r < LN_PPAR; ++r) {
      sg += smem[r * LN_CH + threadIdx.x % LN_CH];
      sb += smem[LN_PPAR * LN_CH + r * LN_CH + threadIdx.x % LN_CH];
    }
    dw[c] = __float2bfloat16(sg);
    db[c] = __float2bfloat16(sb);
  }
}

}  // namespace

std::vector<at::Tensor> layernorm_fwd(at::Tensor x, at::Tensor w, at::Tensor b,
                                      double eps) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
                  w.scalar_type() == at::kBFloat16 &&
                  b.scalar_type() == at::kBFloat16,
              "layernorm: bf16 only");
  const long D = x.size(-1);
  TORCH_CHECK(D % 8 == 0 && D <= 16384, "layernorm: D %8==0, <=16384");
  const long T = x.numel() / D;
  const int c8 = (int)(D / 8);
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto y = at::empty_like(x);
  auto mean = at::empty({T}, x.options().dtype(at::kFloat));
  auto invstd = at::empty({T}, x.options().dtype(at::kFloat));
  hipLaunchKernelGGL(ln_fwd_kernel, dim3((unsigned)T), dim3(LN_BLOCK), 0,
                     stream, (const ln_bf16x8*)x.data_ptr(),
                     (const ln_bf16x8*)w.data_ptr(),
                     (const ln_bf16x8*)b.data_ptr(), (ln_bf16x8*)y.data_ptr(),
                     mean.data_ptr<float>(), invstd.data_ptr<float>(), c8,
                     (int)D, (float)eps);
  return {y, mean, invstd};
}

std::vector<at::Tensor> layernorm_bwd(at::Tensor x, at::Tensor dy,
                                      at::Tensor w, at::Tensor mean,
                                      at::Tensor invstd) {
  const long D = x.size(-1);
  const long T = x.numel() / D;
  const int c8 = (int)(D / 8);
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto dx = at::empty_like(x);
  long target = 1024;
  int rpb = (int)((T + target - 1) / target);
  if (rpb < 1) rpb = 1;
  int nblocks = (int)((T + rpb - 1) / rpb);
  auto part = at::empty({(long)nblocks * 2 * D},
                        x.options().dtype(at::kFloat));
  size_t lds = (size_t)2 * D * sizeof(float);
  hipLaunchKernelGGL(ln_bwd_kernel, dim3(nblocks), dim3(LN_BLOCK), lds,
                     stream, (const ln_bf16x8*)x.data_ptr(),
                     (const ln_bf16x8*)dy.data_ptr(),
                     (const ln_bf16x8*)w.data_ptr(), mean.data_ptr<float>(),
                     invstd.data_ptr<float>(), (ln_bf16x8*)dx.data_ptr(),
                     part.data_ptr<float>(), T, c8, (int)D, rpb);
  auto dw = at::empty_like(w);
  auto db = at::empty_like(w);
  hipLaunchKernelGGL(ln_dwdb_reduce_kernel,
                     dim3((unsigned)((D + LN_CH - 1) / LN_CH)),
                     dim3(LN_PPAR * LN_CH), 0, stream,
                     part.data_ptr<float>(), nblocks, (bf16*)dw.data_ptr(),
                     (bf16*)db.data_ptr(), (int)D);
  return {dx, dw, db};
}
