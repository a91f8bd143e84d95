// Flash-attention backward (FA-2 split) for MI355X gfx950 — v1.
//
// Same 8-wave 32x32-MFMA structure as the v1 forward (csrc/fa_fwd.hip):
// LDS-staged tiles shared by all 8 waves, XOR-swizzled row-major images for
// row-operand (A) reads, padded transposed images for the d-major (A^T)
// reads, and the C->B fragment half-exchange repack (cvt_pk + shfl_xor 32)
// so P / dS never round-trip through LDS.
//
// Math (standard FA-2 with saved logsumexp L and delta = rowsum(dO*O)):
//   P  = exp(S*scale - L)
//   dV += P^T dO        dP = dO V^T
//   dS = P * (dP - delta) * scale
//   dK += dS^T Q        dQ += dS K
//
// Kernel split:
//   fa_bwd_dkv_v1: block owns 256 kv rows (8 waves x 32); loops over the
//     q-heads of its kv head (GQA accumulated IN REGISTERS — no fp32
//     scratch buffers, no host-side reduction) and over 64-row q tiles.
//     Per tile it stages Q and dO row-major (swizzled) AND transposed
//     (padded), computes C[q,kv] products (lane = kv column, so the wave's
//     own K/V registers serve as the B operand), and accumulates
//     dV^T/dK^T[d,kv] with transposed-image A fragments.
//   fa_bwd_dq_v1: block owns 256 q rows; mirrors the forward (K/V staged,
//     Q/dO in registers, C[kv,q] products with lane = q column, dQ^T
//     accumulated from the transposed K image).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

typedef __hip_bfloat16 bf16;
typedef __attribute__((ext_vector_type(8))) short bf16x8v;
typedef __attribute__((ext_vector_type(16))) float f32x16v;

#define FB_QB 64    // q rows staged per dkv tile / kv rows per dq tile
#define FB_PAD 8    // transposed-image row padding (elements)

__device__ __forceinline__ bf16 fb_f2bf(float f) {
  return __float2bfloat16(f);
}

__device__ __forceinline__ float fb_b2f(const bf16& h) {
  return __bfloat162float(h);
}

// pack two f32 -> two bf16 in one dword (one v_cvt_pk_bf16_f32 — the
// scalar cvt+or sequence costs 3-4 VALU ops per dword; guide T12)
__device__ __forceinline__ unsigned int fb_pk(float lo, float hi) {
  unsigned int r;
  asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// Repack one 32-wide C tile (16 f32 regs, rows = reduction dim) into two
// bf16 B/A fragments whose k-elems run along that dim.  Identical to the
// forward's repack: prep both destination halves, one half-exchange.
__device__ __forceinline__ void fb_repack16(const f32x16v& pr, int h2,
                                            bf16x8v out[2]) {
  #pragma unroll
  for (int kc = 0; kc < 2; ++kc) {
    unsigned int d01[2], d23[2];
    #pragma unroll
    for (int ht = 0; ht < 2; ++ht) {
      float pv[4];
      #pragma unroll
      for (int j = 0; j < 4; ++j) {
        // reduction index = kc*16 + ht*8 + h2*4 + j; reg = j + 4*(kc*2+ht)
        // is independent of h2 only through the value selection below
        const int r0 = j + 4 * (kc * 2 + ht);
        pv[j] = pr[r0];
      }
      unsigned int* dst = ht == 0 ? d01 : d23;
      dst[0] = fb_pk(pv[0], pv[1]);
      dst[1] = fb_pk(pv[2], pv[3]);
    }
    const unsigned int s01_0 = __shfl_xor((int)d01[0], 32);
    const unsigned int s01_1 = __shfl_xor((int)d01[1], 32);
    const unsigned int s23_0 = __shfl_xor((int)d23[0], 32);
    const unsigned int s23_1 = __shfl_xor((int)d23[1], 32);
    unsigned int w[4];
    if (h2 == 0) {
      w[0] = d01[0]; w[1] = d01[1]; w[2] = s01_0; w[3] = s01_1;
    } else {
      w[0] = s23_0; w[1] = s23_1; w[2] = d23[0]; w[3] = d23[1];
    }
    out[kc] = *reinterpret_cast<bf16x8v*>(w);
  }
}

// delta[b,h,s] = sum_d dO[b,h,s,d] * O[b,h,s,d]  (fp32)
__global__ __launch_bounds__(256) void fa_delta_kernel(
    const bf16* __restrict__ dout, const bf16* __restrict__ out,
    float* __restrict__ delta, long rows, int DH, int H, int S,
    long dsb, long dsh, long dss, long osb, long osh, long oss) {
  const long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  const int l = threadIdx.x & 63;
  if (row >= rows) return;
  const int sres = (int)(row % S);
  const int hres = (int)((row / S) % H);
  const int bres = (int)(row / ((long)S * H));
  const bf16* dop = dout + (long)bres * dsb + (long)hres * dsh +
                    (long)sres * dss;
  const bf16* op = out + (long)bres * osb + (long)hres * osh +
                   (long)sres * oss;
  float s = 0.f;
  for (int d = l * 2; d < DH; d += 128) {
    s += fb_b2f(dop[d]) * fb_b2f(op[d]);
    s += fb_b2f(dop[d + 1]) * fb_b2f(op[d + 1]);
  }
  #pragma unroll
  for (int off = 32; off > 0; off >>= 1) s += __shfl_down(s, off);
  if (l == 0) delta[row] = s;
}

// ---------------------------------------------------------------------------
// dK/dV: block = 256 kv rows (8 waves x 32); q-head group loop in-register.
// ---------------------------------------------------------------------------
template <int DH, int WAVES>
__global__ __launch_bounds__(WAVES * 64) __attribute__((amdgpu_waves_per_eu(1, 2))) void fa_bwd_dkv_v1(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const bf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    bf16* __restrict__ dk, bf16* __restrict__ dv,
    int S, int H, int HKV, int causal,
    long qsb, long qsh, long qss, long ksb, long ksh, long kss,
    long vsb, long vsh, long vss, long dosb, long dosh, long doss) {
  constexpr int DCH = DH / 16;
  constexpr int DT = DH / 32;
  constexpr int KSLOT = DH / 8;
  constexpr int NT = WAVES * 64;        // threads per block
  constexpr int KVROWS = WAVES * 32;    // kv rows per block
  const int wave = threadIdx.x >> 6;
  const int l = threadIdx.x & 63;
  const int lq = l & 31;
  const int h2 = l >> 5;
  const int bh = blockIdx.y;            // over B * HKV
  const int hkv = bh % HKV;
  const int b = bh / HKV;
  const int G = H / HKV;
  // Causal load-balancing: the host launches HALF the kv strips and each
  // block processes strip b AND its mirror (nstrip-1-b) back to back, so
  // every block sweeps ~the same number of q tiles (a kv=0 block alone
  // sweeps 32x more than a kv=max one — PMC showed the tail running at
  // <10% concurrency on the naive one-strip-per-block grid).
  const int nstrip = (S + KVROWS - 1) / KVROWS;
  const bf16* kp = k + (long)b * ksb + (long)hkv * ksh;
  const bf16* vp = v + (long)b * vsb + (long)hkv * vsh;

  for (int halfi = 0; halfi < 2; ++halfi) {
    const int strip = halfi == 0 ? (int)blockIdx.x
                                 : nstrip - 1 - (int)blockIdx.x;
    if (halfi == 1 && (!causal || strip <= (int)blockIdx.x)) break;
    const int kvbase = strip * KVROWS;
    const int kvw0 = kvbase + wave * 32;           // wave's first kv row
    const int kvrow = min(kvw0 + lq, S - 1);       // lane's kv row (clamped)

  // Wave-private K/V fragments (the 32x32x16 A and B lane maps are
  // identical, so these registers serve as the B operand directly).
  bf16x8v kf[DCH], vf[DCH];
  #pragma unroll
  for (int ch = 0; ch < DCH; ++ch) {
    kf[ch] = *reinterpret_cast<const bf16x8v*>(
        kp + (long)kvrow * kss + ch * 16 + h2 * 8);
    vf[ch] = *reinterpret_cast<const bf16x8v*>(
        vp + (long)kvrow * vss + ch * 16 + h2 * 8);
  }

  f32x16v dkacc[DT], dvacc[DT];
  #pragma unroll
  for (int dt = 0; dt < DT; ++dt)
    #pragma unroll
    for (int r = 0; r < 16; ++r) {
      dkacc[dt][r] = 0.f;
      dvacc[dt][r] = 0.f;
    }

  // LDS images of the current q tile
  __shared__ bf16 qrm[FB_QB * DH];                 // row-major, swizzled
  __shared__ bf16 dorm[FB_QB * DH];
  __shared__ bf16 qtr[DH * (FB_QB + FB_PAD)];      // transposed, padded
  __shared__ bf16 dotr[DH * (FB_QB + FB_PAD)];
  __shared__ float lsh[FB_QB], dsh[FB_QB];

  const float scale = rsqrtf((float)DH);            // natural (dS math)
  const float scale2 = scale * 1.44269504089f;      // exp2-domain exponent
  const int tid = threadIdx.x;

  // ---- tile pipeline (T14): the row-major Q/dO pieces for tile t+1 are
  // loaded to registers during tile t's compute; the transposed images are
  // built from the row-major LDS images (halving global traffic).  Three
  // barriers per tile: [write rm] | [build tr from rm] | [compute].
  constexpr int NPREF = 2 * FB_QB * KSLOT / NT;  // rm pieces per thread
  bf16x8v pref[NPREF];

  auto load_rm = [&](const bf16* qp, const bf16* dop, long qss_, long doss_,
                     int qb) {
    #pragma unroll
    for (int i = 0; i < NPREF; ++i) {
      const int p = tid + i * NT;
      const int img = p >= FB_QB * KSLOT;
      const int pp = p - img * FB_QB * KSLOT;
      const int row = pp / KSLOT;
      const int slot = pp % KSLOT;
      const int grow = min(qb + row, S - 1);
      const bf16* src = img == 0 ? qp : dop;
      const long rs = img == 0 ? qss_ : doss_;
      pref[i] = *reinterpret_cast<const bf16x8v*>(
          src + (long)grow * rs + slot * 8);
    }
  };

  auto write_rm = [&]() {
    #pragma unroll
    for (int i = 0; i < NPREF; ++i) {
      const int p = tid + i * NT;
      const int img = p >= FB_QB * KSLOT;
      const int pp = p - img * FB_QB * KSLOT;
      const int row = pp / KSLOT;
      const int slot = pp % KSLOT;
      const int sslot = slot ^ (row & (KSLOT - 1));
      bf16* dst = img == 0 ? qrm : dorm;
      *reinterpret_cast<bf16x8v*>(&dst[row * DH + sslot * 8]) = pref[i];
    }
  };

  auto build_tr = [&]() {
    for (int a = tid; a < 2 * (DH / 8) * 16; a += NT) {
      const int img = a >= (DH / 8) * 16;
      const int aa = a - img * (DH / 8) * 16;
      const int dchunk = aa >> 4;
      const int rq = aa & 15;             // 4-row group of q rows
      const bf16* srcm = img == 0 ? qrm : dorm;
      bf16* dst = img == 0 ? qtr : dotr;
      union { bf16x8v v8[4]; short sh[4][8]; } u;
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = rq * 4 + i;
        const int sslot = dchunk ^ (row & (KSLOT - 1));
        u.v8[i] = *reinterpret_cast<const bf16x8v*>(
            &srcm[row * DH + sslot * 8]);
      }
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        union { short s4[4]; unsigned long long d; } pack;
        #pragma unroll
        for (int i = 0; i < 4; ++i) pack.s4[i] = u.sh[i][j];
        *reinterpret_cast<unsigned long long*>(
            &dst[(dchunk * 8 + j) * (FB_QB + FB_PAD) + rq * 4]) = pack.d;
      }
    }
  };

  const int q0 = causal ? min(kvbase & ~(FB_QB - 1), S) : 0;
  {
    const int h0 = hkv * G;
    load_rm(q + (long)b * qsb + (long)h0 * qsh,
            dout + (long)b * dosb + (long)h0 * dosh, qss, doss, q0);
  }
  write_rm();
  __syncthreads();
  build_tr();

  for (int gh = 0; gh < G; ++gh) {
    const int h = hkv * G + gh;
    const bf16* qp = q + (long)b * qsb + (long)h * qsh;
    const bf16* dop = dout + (long)b * dosb + (long)h * dosh;
    const float* lp = lse + ((long)b * H + h) * S;
    const float* dp = delta + ((long)b * H + h) * S;

    for (int qb = q0; qb < S; qb += FB_QB) {
      if (tid < FB_QB) {
        const int grow = min(qb + tid, S - 1);
        lsh[tid] = lp[grow] * 1.44269504089f;  // logsumexp in base-2 units
        dsh[tid] = dp[grow];
      }
      __syncthreads();

      // next tile in the (gh, qb) sequence, for the prefetch
      const int qb_n = qb + FB_QB < S ? qb + FB_QB : q0;
      const int gh_n = qb + FB_QB < S ? gh : gh + 1;
      const bool has_next = gh_n < G;

      // ---- per 32-row q sub-tile: C[q, kv] products, elementwise, repack,
      // dV^T/dK^T accumulate
      #pragma unroll
      for (int ct = 0; ct < 2; ++ct) {
        f32x16v sacc, dpacc;
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          sacc[r] = 0.f;
          dpacc[r] = 0.f;
        }
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int ch = 0; ch < DCH; ++ch) {
          const int row = ct * 32 + lq;
          const int slot = (ch * 2 + h2) ^ (row & (KSLOT - 1));
          bf16x8v qa = *reinterpret_cast<const bf16x8v*>(
              &qrm[row * DH + slot * 8]);
          bf16x8v doa = *reinterpret_cast<const bf16x8v*>(
              &dorm[row * DH + slot * 8]);
          sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qa, kf[ch], sacc,
                                                         0, 0, 0);
          dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(doa, vf[ch], dpacc,
                                                          0, 0, 0);
        }
        __builtin_amdgcn_s_setprio(0);
        if (ct == 0 && has_next) {
          // issue next tile's global loads under the remaining compute
          const int hh = hkv * G + gh_n;
          load_rm(q + (long)b * qsb + (long)hh * qsh,
                  dout + (long)b * dosb + (long)hh * dosh, qss, doss, qb_n);
        }
        // overwrite sacc/dpacc in place with P / dS (register economy)
        const int kvg = kvw0 + lq;                 // lane's kv column
        #pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int qloc = ct * 32 + (r & 3) + 8 * (r >> 2) + 4 * h2;
          const int qg = qb + qloc;
          float p = 0.f;
          if (qg < S && kvg < S && (!causal || kvg <= qg))
            p = __builtin_amdgcn_exp2f(sacc[r] * scale2 - lsh[qloc]);
          const float ds = p * (dpacc[r] - dsh[qloc]) * scale;
          sacc[r] = p;
          dpacc[r] = ds;
        }
        bf16x8v pb[2], dsb[2];
        fb_repack16(sacc, h2, pb);
        fb_repack16(dpacc, h2, dsb);
        __builtin_amdgcn_s_setprio(1);
        #pragma unroll
        for (int dt = 0; dt < DT; ++dt) {
          #pragma unroll
          for (int kc = 0; kc < 2; ++kc) {
            const int qc = (ct * 2 + kc) * 16 + h2 * 8;  // q k-chunk base
            bf16x8v dota = *reinterpret_cast<const bf16x8v*>(
                &dotr[(dt * 32 + lq) * (FB_QB + FB_PAD) + qc]);
            bf16x8v qta = *reinterpret_cast<const bf16x8v*>(
                &qtr[(dt * 32 + lq) * (FB_QB + FB_PAD) + qc]);
            dvacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                dota, pb[kc], dvacc[dt], 0, 0, 0);
            dkacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                qta, dsb[kc], dkacc[dt], 0, 0, 0);
          }
        }
        __builtin_amdgcn_s_setprio(0);
      }
      __syncthreads();              // everyone done reading rm + tr
      if (has_next) {
        write_rm();
        __syncthreads();            // rm image of tile t+1 complete
        build_tr();
      }
    }
  }

  // ---- epilogue: C[d, kv] (lane = kv column) -> dk/dv[kv][d] bf16
  const int kvg = kvw0 + lq;
  if (kvg < S) {
    bf16* dkp = dk + (((long)b * HKV + hkv) * S + kvg) * DH;
    bf16* dvp = dv + (((long)b * HKV + hkv) * S + kvg) * DH;
    // (dk/dv are allocated contiguous [B,HKV,S,D] by the host wrapper)
    #pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      #pragma unroll
      for (int r = 0; r < 16; r += 2) {
        const int d = dt * 32 + (r & 3) + 8 * (r >> 2) + 4 * h2;
        *reinterpret_cast<unsigned int*>(dkp + d) =
            fb_pk(dkacc[dt][r], dkacc[dt][r + 1]);
        *reinterpret_cast<unsigned int*>(dvp + d) =
            fb_pk(dvacc[dt][r], dvacc[dt][r + 1]);
      }
    }
  }
  __syncthreads();  // LDS images reused by the mirror strip
  }  // halfi
}

// ---------------------------------------------------------------------------
// dQ: block = 256 q rows (8 waves x 32); mirrors the forward's structure.
// ---------------------------------------------------------------------------
template <int DH>
__global__ __launch_bounds__(512) void fa_bwd_dq_v1(
    const bf16* __restrict__ q, const bf16* __restrict__ k,
    const bf16* __restrict__ v, const bf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    bf16* __restrict__ dq, int S, int H, int HKV, int causal,
    long qsb, long qsh, long qss, long ksb, long ksh, long kss,
    long vsb, long vsh, long vss, long dosb, long dosh, long doss) {
  constexpr int DCH = DH / 16;
  constexpr int DT = DH / 32;
  constexpr int KSLOT = DH / 8;
  const int wave = threadIdx.x >> 6;
  const int l = threadIdx.x & 63;
  const int lq = l & 31;
  const int h2 = l >> 5;
  const int bh = blockIdx.y;
  const int h = bh % H;
  const int hkv = h / (H / HKV);
  const int b = bh / H;
  // Causal load balance: block handles q strip b AND its mirror (cf. dkv)
  const int nstrip = (S + 255) / 256;
  const bf16* qp = q + (long)b * qsb + (long)h * qsh;
  const bf16* kp = k + (long)b * ksb + (long)hkv * ksh;
  const bf16* vp = v + (long)b * vsb + (long)hkv * vsh;
  const bf16* dop = dout + (long)b * dosb + (long)h * dosh;

  for (int halfi = 0; halfi < 2; ++halfi) {
  const int strip = halfi == 0 ? (int)blockIdx.x
                               : nstrip - 1 - (int)blockIdx.x;
  if (halfi == 1 && (!causal || strip <= (int)blockIdx.x)) break;
  const int qbase = strip * 256;
  const int row0 = qbase + wave * 32;
  const int qrow = min(row0 + lq, S - 1);
  const float Lq = lse[((long)b * H + h) * S + qrow] * 1.44269504089f;
  const float Dq = delta[((long)b * H + h) * S + qrow];

  bf16x8v qf[DCH], dof[DCH];
  #pragma unroll
  for (int ch = 0; ch < DCH; ++ch) {
    qf[ch] = *reinterpret_cast<const bf16x8v*>(
        qp + (long)qrow * qss + ch * 16 + h2 * 8);
    dof[ch] = *reinterpret_cast<const bf16x8v*>(
        dop + (long)qrow * doss + ch * 16 + h2 * 8);
  }
  f32x16v dqacc[DT];
  #pragma unroll
  for (int dt = 0; dt < DT; ++dt)
    #pragma unroll
    for (int r = 0; r < 16; ++r) dqacc[dt][r] = 0.f;

  __shared__ bf16 krm[FB_QB * DH];
  __shared__ bf16 vrm[FB_QB * DH];
  __shared__ bf16 ktr[DH * (FB_QB + FB_PAD)];

  const float scale = rsqrtf((float)DH);            // natural (dS math)
  const float scale2 = scale * 1.44269504089f;      // exp2-domain exponent
  const int tid = threadIdx.x;

  // ---- tile pipeline (cf. fa_bwd_dkv_v1): K/V row-major pieces for tile
  // t+1 prefetched to registers during tile t's compute; the transposed K
  // image is built from the row-major LDS image.
  constexpr int NT = 512;
  constexpr int NPREF = 2 * FB_QB * KSLOT / NT;
  bf16x8v pref[NPREF];

  auto load_rm = [&](int kb) {
    #pragma unroll
    for (int i = 0; i < NPREF; ++i) {
      const int p = tid + i * NT;
      const int img = p >= FB_QB * KSLOT;
      const int pp = p - img * FB_QB * KSLOT;
      const int row = pp / KSLOT;
      const int slot = pp % KSLOT;
      const int grow = min(kb + row, S - 1);
      const bf16* src = img == 0 ? kp : vp;
      const long rs = img == 0 ? kss : vss;
      pref[i] = *reinterpret_cast<const bf16x8v*>(
          src + (long)grow * rs + slot * 8);
    }
  };

  auto write_rm = [&]() {
    #pragma unroll
    for (int i = 0; i < NPREF; ++i) {
      const int p = tid + i * NT;
      const int img = p >= FB_QB * KSLOT;
      const int pp = p - img * FB_QB * KSLOT;
      const int row = pp / KSLOT;
      const int slot = pp % KSLOT;
      const int sslot = slot ^ (row & (KSLOT - 1));
      bf16* dst = img == 0 ? krm : vrm;
      *reinterpret_cast<bf16x8v*>(&dst[row * DH + sslot * 8]) = pref[i];
    }
  };

  auto build_tr = [&]() {
    for (int a = tid; a < (DH / 8) * 16; a += NT) {
      const int dchunk = a >> 4;
      const int rq = a & 15;
      union { bf16x8v v8[4]; short sh[4][8]; } u;
      #pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int row = rq * 4 + i;
        const int sslot = dchunk ^ (row & (KSLOT - 1));
        u.v8[i] = *reinterpret_cast<const bf16x8v*>(
            &krm[row * DH + sslot * 8]);
      }
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        union { short s4[4]; unsigned long long d; } pack;
        #pragma unroll
        for (int i = 0; i < 4; ++i) pack.s4[i] = u.sh[i][j];
        *reinterpret_cast<unsigned long long*>(
            &ktr[(dchunk * 8 + j) * (FB_QB + FB_PAD) + rq * 4]) = pack.d;
      }
    }
  };

  const int kv_end = causal ? min(qbase + 256, S) : S;
  load_rm(0);
  write_rm();
  __syncthreads();
  build_tr();
  __syncthreads();
  for (int kb = 0; kb < kv_end; kb += FB_QB) {
    #pragma unroll
    for (int ct = 0; ct < 2; ++ct) {
      f32x16v sacc, dpacc;
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        sacc[r] = 0.f;
        dpacc[r] = 0.f;
      }
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int ch = 0; ch < DCH; ++ch) {
        const int row = ct * 32 + lq;
        const int slot = (ch * 2 + h2) ^ (row & (KSLOT - 1));
        bf16x8v kfr = *reinterpret_cast<const bf16x8v*>(
            &krm[row * DH + slot * 8]);
        bf16x8v vfr = *reinterpret_cast<const bf16x8v*>(
            &vrm[row * DH + slot * 8]);
        sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfr, qf[ch], sacc,
                                                       0, 0, 0);
        dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfr, dof[ch], dpacc,
                                                        0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
      // C[kv, q]: lane = q column; elementwise uses the lane's own L/delta
      const int qg = row0 + lq;
      #pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int kvg = kb + ct * 32 + (r & 3) + 8 * (r >> 2) + 4 * h2;
        float p = 0.f;
        if (kvg < S && qg < S && (!causal || kvg <= qg))
          p = __builtin_amdgcn_exp2f(sacc[r] * scale2 - Lq);
        dpacc[r] = p * (dpacc[r] - Dq) * scale;
      }
      bf16x8v dsb[2];
      fb_repack16(dpacc, h2, dsb);
      if (ct == 0 && kb + FB_QB < kv_end)
        load_rm(kb + FB_QB);  // overlap next tile's HBM latency
      __builtin_amdgcn_s_setprio(1);
      #pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        #pragma unroll
        for (int kc = 0; kc < 2; ++kc) {
          const int kvc = (ct * 2 + kc) * 16 + h2 * 8;
          bf16x8v kta = *reinterpret_cast<const bf16x8v*>(
              &ktr[(dt * 32 + lq) * (FB_QB + FB_PAD) + kvc]);
          dqacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              kta, dsb[kc], dqacc[dt], 0, 0, 0);
        }
      }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
    if (kb + FB_QB < kv_end) {
      write_rm();
      __syncthreads();
      build_tr();
      __syncthreads();
    }
  }

  // epilogue: dQ^T[d, q] (lane = q) -> dq[q][d]
  const int qg = row0 + lq;
  if (qg < S) {
    bf16* dqp = dq + (((long)b * H + h) * S + qg) * DH;
    #pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      #pragma unroll
      for (int r = 0; r < 16; r += 2) {
        const int d = dt * 32 + (r & 3) + 8 * (r >> 2) + 4 * h2;
        *reinterpret_cast<unsigned int*>(dqp + d) =
            fb_pk(dqacc[dt][r], dqacc[dt][r + 1]);
      }
    }
  }
  __syncthreads();  // LDS images reused by the mirror strip
  }  // halfi
}

}  // namespace

static bool fb_strides_ok(const at::Tensor& t) {
  return t.stride(3) == 1 && t.stride(0) % 8 == 0 && t.stride(1) % 8 == 0 &&
         t.stride(2) % 8 == 0;
}

std::vector<at::Tensor> fa_bwd(at::Tensor q, at::Tensor k, at::Tensor v,
                               at::Tensor out, at::Tensor dout,
                               at::Tensor lse, bool causal) {
  if (!fb_strides_ok(q)) q = q.contiguous();
  if (!fb_strides_ok(k)) k = k.contiguous();
  if (!fb_strides_ok(v)) v = v.contiguous();
  if (!fb_strides_ok(out)) out = out.contiguous();
  if (!fb_strides_ok(dout)) dout = dout.contiguous();
  const int B = q.size(0), H = q.size(1), S = q.size(2), DH = q.size(3);
  const int HKV = k.size(1);
  TORCH_CHECK(DH == 64 || DH == 128, "fa_bwd: head dim 64 or 128");
  auto stream = at::hip::getCurrentHIPStream().stream();
  auto delta = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  const long rows = (long)B * H * S;
  hipLaunchKernelGGL(fa_delta_kernel, dim3((unsigned)((rows + 3) / 4)),
                     dim3(256), 0, stream, (const bf16*)dout.data_ptr(),
                     (const bf16*)out.data_ptr(), delta.data_ptr<float>(),
                     rows, DH, H, S,
                     dout.stride(0), dout.stride(1), dout.stride(2),
                     out.stride(0), out.stride(1), out.stride(2));
  // grads allocated contiguous [B,H,S,D]/[B,HKV,S,D] (autograd accepts
  // any layout; the kernels' write paths assume dense BHSD)
  auto dq = at::empty({B, H, S, DH}, q.options());
  auto dk = at::empty({B, HKV, S, DH}, q.options());
  auto dv = at::empty({B, HKV, S, DH}, q.options());
  // D=128 dkv runs 4 waves/block (whole VGPR file per wave: spill-free).
  // Causal runs launch half the kv strips; each block also does its mirror.
  const int ns128 = (S + 127) / 128, ns256 = (S + 255) / 256;
  dim3 grid_kv128(causal ? (ns128 + 1) / 2 : ns128, B * HKV);
  dim3 grid_kv(causal ? (ns256 + 1) / 2 : ns256, B * HKV);
  const int nsq = (S + 255) / 256;
  dim3 grid_q(causal ? (nsq + 1) / 2 : nsq, B * H);
  const auto LKV = [&](auto kern, dim3 g, dim3 blk) {
    hipLaunchKernelGGL(kern, g, blk, 0, stream,
                       (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                       (const bf16*)v.data_ptr(), (const bf16*)dout.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       (bf16*)dk.data_ptr(), (bf16*)dv.data_ptr(),
                       S, H, HKV, causal ? 1 : 0,
                       q.stride(0), q.stride(1), q.stride(2),
                       k.stride(0), k.stride(1), k.stride(2),
                       v.stride(0), v.stride(1), v.stride(2),
                       dout.stride(0), dout.stride(1), dout.stride(2));
  };
  const auto LQ = [&](auto kern) {
    hipLaunchKernelGGL(kern, grid_q, dim3(512), 0, stream,
                       (const bf16*)q.data_ptr(), (const bf16*)k.data_ptr(),
                       (const bf16*)v.data_ptr(), (const bf16*)dout.data_ptr(),
                       lse.data_ptr<float>(), delta.data_ptr<float>(),
                       (bf16*)dq.data_ptr(), S, H, HKV, causal ? 1 : 0,
                       q.stride(0), q.stride(1), q.stride(2),
                       k.stride(0), k.stride(1), k.stride(2),
                       v.stride(0), v.stride(1), v.stride(2),
                       dout.stride(0), dout.stride(1), dout.stride(2));
  };
  if (DH == 128) {
    LKV(fa_bwd_dkv_v1<128, 4>, grid_kv128, dim3(256));
    LQ(fa_bwd_dq_v1<128>);
  } else {
    LKV(fa_bwd_dkv_v1<64, 8>, grid_kv, dim3(512));
    LQ(fa_bwd_dq_v1<64>);
  }
  return {dq, dk, dv};
}
