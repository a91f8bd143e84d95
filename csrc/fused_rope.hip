// Fused rotary position embedding (RoPE, interleaved-pair convention) for
// MI355X.  The eager composition (strided even/odd slicing, 6+ kernels and
// two scatter writes per projection) runs once per q and k per layer; this
// kernel does the rotation in one vectorized streaming pass on the
// CONTIGUOUS [B, S, H, Dh] projection (before the attention transpose).
//
// backward = the same rotation with sin negated (rotation transpose), so one
// kernel serves both directions via `conj`.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define RP_BLOCK 256

namespace {

typedef __hip_bfloat16 bf16;

union rp_bf16x8 {
  uint4 u4;
  unsigned short h[8];
};

__device__ __forceinline__ float rp_b2f(unsigned short v) {
  __hip_bfloat16_raw r;
  r.x = v;
  return __bfloat162float(*reinterpret_cast<bf16*>(&r));
}

__device__ __forceinline__ unsigned short rp_f2b(float f) {
  bf16 h = __float2bfloat16(f);
  return *reinterpret_cast<unsigned short*>(&h);
}

// x: [B, S, H, Dh] bf16 contiguous viewed as octets; cos/sin: [>=S, Dh/2]
// fp32.  Octet `oct` within a head covers head-dim positions 8*oct..8*oct+7
// = interleaved pairs (x[2i], x[2i+1]) for i = 4*oct..4*oct+3.
template <bool CONJ>
__global__ __launch_bounds__(RP_BLOCK) void rope_kernel(
    const rp_bf16x8* __restrict__ x, rp_bf16x8* __restrict__ y,
    const float* __restrict__ cosc, const float* __restrict__ sinc,
    long n8, int h8, int H, int S, int half) {
  long stride = (long)gridDim.x * RP_BLOCK;
  for (long i = (long)blockIdx.x * RP_BLOCK + threadIdx.x; i < n8;
       i += stride) {
    const int oct = (int)(i % h8);
    const long row = i / h8;        // (b*S + s)*H + h
    const int s = (int)((row / H) % S);
    const float* crow = cosc + (long)s * half + oct * 4;
    const float* srow = sinc + (long)s * half + oct * 4;
    const rp_bf16x8 v = x[i];
    rp_bf16x8 o;
    #pragma unroll
    for (int p = 0; p < 4; ++p) {
      const float c = crow[p];
      const float sn = CONJ ? -srow[p] : srow[p];
      const float x1 = rp_b2f(v.h[2 * p]);
      const float x2 = rp_b2f(v.h[2 * p + 1]);
      o.h[2 * p] = rp_f2b(x1 * c - x2 * sn);
      o.h[2 * p + 1] = rp_f2b(x2 * c + x1 * sn);
    }
    y[i] = o;
  }
}

}  // namespace

// x: [B, S, H, Dh] bf16 contiguous; cos/sin: [>=S, Dh/2] fp32 contiguous.
at::Tensor rope_apply(at::Tensor x, at::Tensor cosc, at::Tensor sinc,
                      bool conj) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.dim() == 4 &&
                  x.is_contiguous(),
              "rope: contiguous 4-D bf16 input required");
  const int B = (int)x.size(0), S = (int)x.size(1), H = (int)x.size(2),
            Dh = (int)x.size(3);
  TORCH_CHECK(Dh % 8 == 0, "rope: head dim must be a multiple of 8");
  TORCH_CHECK(cosc.size(-1) == Dh / 2 && cosc.size(0) >= S,
              "rope: cos cache shape mismatch");
  auto y = at::empty_like(x);
  const int h8 = Dh / 8;
  const long n8 = (long)B * S * H * h8;
  auto stream = at::hip::getCurrentHIPStream().stream();
  int grid = (int)std::min<long>((n8 + RP_BLOCK - 1) / RP_BLOCK, 4096);
  if (conj) {
    hipLaunchKernelGGL((rope_kernel<true>), dim3(grid), dim3(RP_BLOCK), 0,
                       stream, (const rp_bf16x8*)x.data_ptr(),
                       (rp_bf16x8*)y.data_ptr(), cosc.data_ptr<float>(),
                       sinc.data_ptr<float>(), n8, h8, H, S, Dh / 2);
  } else {
    hipLaunchKernelGGL((rope_kernel<false>), dim3(grid), dim3(RP_BLOCK), 0,
                       stream, (const rp_bf16x8*)x.data_ptr(),
                       (rp_bf16x8*)y.data_ptr(), cosc.data_ptr<float>(),
                       sinc.data_ptr<float>(), n8, h8, H, S, Dh / 2);
  }
  return y;
}
