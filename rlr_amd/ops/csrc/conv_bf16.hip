// bf16 implicit-GEMM convolution on MFMA (v_mfma_f32_16x16x32_bf16) —
// NHWC, fp32 accumulate; the compute path for --dtype bf16 (BASELINE
// config 3: CIFAR10 ResNet18 bf16).
//
// Same tiling discipline as conv_f32.hip; K-step = the MFMA k-depth (32),
// so each staged tile feeds exactly one MFMA per subtile.  Both operands
// whose fragment runs along K are staged TRANSPOSED so a fragment read is
// one 16-byte load of 8 contiguous bf16.
//
// Supported fast paths (the binding falls back to the fp32 kernels with
// casts outside them): fwd C % 32 == 0; bwd-data KO % 32 == 0 and
// C % 4 == 0; bwd-weight C % 8 == 0.  First layers (C=1,3) take the fp32
// fallback — they are a rounding error of ResNet's FLOPs.
#include "common.h"

typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef short bf16x8 __attribute__((ext_vector_type(8)));

constexpr int BKB = 32;           // K-step (bf16 MFMA k-depth)
constexpr int LDA_B = BKB + 8;    // bf16 elements per A_lds row
constexpr int LDT_B = BKB + 8;    // transposed images: [col][k] rows

struct ConvShapeB {
  int Nb, C, H, W, Kout, R, S, OH, OW, stride, pad;
};

// ------------------------------------------------------------------- fwd

template <bool P0>
__global__ __launch_bounds__(256)
void conv_fwd_bf16_k(const unsigned short* __restrict__ x,
                     const unsigned short* __restrict__ wt,  // [(r,s,c)][KO]
                     const float* __restrict__ bias,
                     unsigned short* __restrict__ y, ConvShapeB sh,
                     int Kdim, int relu) {
  constexpr int BM = 128, BN = 64, MI = 4, NI = 2;
  __shared__ unsigned short A_lds[2][BM * LDA_B];
  __shared__ unsigned short B_lds[2][BN * LDT_B];  // transposed [ko][k]
  const int t = threadIdx.x;
  const int wave = t >> 6, lane = t & 63;
  const int wr = wave >> 1, wc = wave & 1;
  const int l15 = lane & 15, l4 = lane >> 4;
  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  const int m_blk = blockIdx.x * BM;
  const int n_blk = blockIdx.y * BN;
  const long M = (long)sh.Nb * sh.OH * sh.OW;
  // A: 128x32 bf16 / 256 threads = 2 x (8 bf16); row = t>>2 (+64), k=(t&3)*8
  const int am = t >> 2, ak = (t & 3) * 8;
  // B: load wt row gk (8 contiguous ko), write transposed
  const int bko = (t & 7) * 8, bkr = t >> 3;

  int ow0[2], oh0[2];
  long base[2];  // patch base incl. tap 0 + ak: gather = base[j] + tap_off
  bool mval[2];
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    long gm = m_blk + am + j * 64;
    mval[j] = gm < M;
    long gmc = mval[j] ? gm : 0;
    ow0[j] = (int)(gmc % sh.OW) * sh.stride - sh.pad;
    oh0[j] = (int)((gmc / sh.OW) % sh.OH) * sh.stride - sh.pad;
    base[j] = (gmc / ((long)sh.OW * sh.OH)) * (long)sh.H * sh.W * sh.C +
              ((long)oh0[j] * sh.W + ow0[j]) * sh.C + ak;
  }

  // staged-tap state carried incrementally across stage_load calls
  // (ascending k0; C % 32 == 0 on this path) — no div/mod in the loop
  int tr = 0, ts = 0, tcb = 0;
  long tap_off = 0;
  const bool wt_v8 = (sh.Kout % 8) == 0 && n_blk + bko + 7 < sh.Kout;
  long wt_base = (long)bkr * sh.Kout + n_blk + bko;

  unsigned short ra[2][8], rb[8];
  auto stage_load = [&](int k0) {
    const int r = tr, s = ts;
    const long toff = tap_off;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      bf16x8 q = {0, 0, 0, 0, 0, 0, 0, 0};
      bool ok = mval[j];
      if (!P0) {
        int ih = oh0[j] + r, iw = ow0[j] + s;
        ok = ok && (unsigned)ih < (unsigned)sh.H &&
             (unsigned)iw < (unsigned)sh.W;
      }
      if (ok) q = *(const bf16x8*)(x + base[j] + toff);
      *(bf16x8*)ra[j] = q;
    }
    tcb += BKB; tap_off += BKB;
    if (tcb == sh.C) {
      tcb = 0;
      if (++ts == sh.S) { ts = 0; ++tr;
                          tap_off += (long)(sh.W - sh.S) * sh.C; }
    }
    {
      int gk = k0 + bkr;
      bf16x8 q = {0, 0, 0, 0, 0, 0, 0, 0};
      if (gk < Kdim) {
        if (wt_v8)
          q = *(const bf16x8*)(wt + wt_base);
        else {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            if (n_blk + bko + e < sh.Kout)
              ((unsigned short*)&q)[e] = wt[wt_base + e];
        }
      }
      *(bf16x8*)rb = q;
    }
    wt_base += (long)BKB * sh.Kout;
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int j = 0; j < 2; ++j)
      *(bf16x8*)&A_lds[buf][(am + j * 64) * LDA_B + ak] = *(bf16x8*)ra[j];
#pragma unroll
    for (int e = 0; e < 8; ++e)
      B_lds[buf][(bko + e) * LDT_B + bkr] = rb[e];
  };

  stage_load(0);
  stage_write(0);
  if (BKB < Kdim) stage_load(BKB);
  __syncthreads();
  int buf = 0;
  for (int k0 = 0; k0 < Kdim; k0 += BKB) {
    if (k0 + BKB < Kdim) {
      stage_write(buf ^ 1);
      if (k0 + 2 * BKB < Kdim) stage_load(k0 + 2 * BKB);
    }
    const unsigned short* Ab = A_lds[buf];
    const unsigned short* Bb = B_lds[buf];
    bf16x8 a_frag[MI], b_frag[NI];
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
      a_frag[mi] = *(const bf16x8*)&Ab[(wr * 64 + mi * 16 + l15) * LDA_B +
                                       l4 * 8];
#pragma unroll
    for (int ni = 0; ni < NI; ++ni)
      b_frag[ni] = *(const bf16x8*)&Bb[(wc * 32 + ni * 16 + l15) * LDT_B +
                                       l4 * 8];
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
#pragma unroll
      for (int ni = 0; ni < NI; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    __syncthreads();
    buf ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      int ko = n_blk + wc * 32 + ni * 16 + l15;
      if (ko >= sh.Kout) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long m = m_blk + wr * 64 + mi * 16 + l4 * 4 + r;
        if (m >= M) continue;
        float v = acc[mi][ni][r];
        if (bias) v += bias[ko];
        if (relu) v = fmaxf(v, 0.f);
        y[m * sh.Kout + ko] = f2bf_(v);
      }
    }
}

// -------------------------------------------------------------- bwd-data

template <int ST>
__global__ __launch_bounds__(256)
void conv_bwd_data_bf16_k(const unsigned short* __restrict__ dy,
                          const unsigned short* __restrict__ wp,  // [(r,s,ko)][C]
                          unsigned short* __restrict__ dx, ConvShapeB sh,
                          int Kdim,
                          const unsigned short* __restrict__ relu_y) {
  constexpr int BM = 128, BN = 64, MI = 4, NI = 2;
  __shared__ unsigned short A_lds[2][BM * LDA_B];
  __shared__ unsigned short B_lds[2][BN * LDT_B];  // transposed [c][k]
  const int t = threadIdx.x;
  const int wave = t >> 6, lane = t & 63;
  const int wr = wave >> 1, wc = wave & 1;
  const int l15 = lane & 15, l4 = lane >> 4;
  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  const int m_blk = blockIdx.x * BM;
  const int n_blk = blockIdx.y * BN;
  const long M = (long)sh.Nb * sh.H * sh.W;
  const int am = t >> 2, ak = (t & 3) * 8;
  const int bc8 = (t & 7) * 8, bkr = t >> 3;
  const int stride = ST > 0 ? ST : sh.stride;

  int iwp[2], ihp[2];
  long base[2];  // ST==1: folded patch base (gather = base[j] - tile_off)
  bool mval[2];
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    long gm = m_blk + am + j * 64;
    mval[j] = gm < M;
    long gmc = mval[j] ? gm : 0;
    iwp[j] = (int)(gmc % sh.W) + sh.pad;
    ihp[j] = (int)((gmc / sh.W) % sh.H) + sh.pad;
    base[j] = (gmc / ((long)sh.W * sh.H)) * (long)sh.OH * sh.OW * sh.Kout;
    if (ST == 1)
      base[j] += ((long)ihp[j] * sh.OW + iwp[j]) * sh.Kout + ak;
  }

  // staged-tap state (ascending k0; KO % 32 == 0 on this path)
  int tr = 0, ts = 0, tkb = 0;
  long noff = 0;
  const bool wp_v8 = (sh.C % 8) == 0 && n_blk + bc8 + 7 < sh.C;
  long wp_base = (long)bkr * sh.C + n_blk + bc8;

  unsigned short ra[2][8], rb[8];
  auto stage_load = [&](int k0) {
    const int r = tr, s = ts;
    if (ST == 1) {
      const long toff = noff;
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        int ohn = ihp[j] - r, own = iwp[j] - s;
        bf16x8 q = {0, 0, 0, 0, 0, 0, 0, 0};
        if (mval[j] && (unsigned)ohn < (unsigned)sh.OH &&
            (unsigned)own < (unsigned)sh.OW)
          q = *(const bf16x8*)(dy + base[j] - toff);
        *(bf16x8*)ra[j] = q;
      }
    } else {
      const int ko0 = tkb + ak;
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        int ohn = ihp[j] - r, own = iwp[j] - s;
        bf16x8 q = {0, 0, 0, 0, 0, 0, 0, 0};
        if (mval[j] && ohn >= 0 && own >= 0 && ohn % stride == 0 &&
            own % stride == 0) {
          int oh = ohn / stride, ow = own / stride;
          if (oh < sh.OH && ow < sh.OW)
            q = *(const bf16x8*)(dy + base[j] +
                                 ((long)oh * sh.OW + ow) * sh.Kout + ko0);
        }
        *(bf16x8*)ra[j] = q;
      }
    }
    tkb += BKB; noff -= BKB;
    if (tkb == sh.Kout) {
      tkb = 0; noff += 2L * sh.Kout;
      if (++ts == sh.S) { ts = 0; ++tr;
                          noff += (long)(sh.OW - sh.S) * sh.Kout; }
    }
    {
      int gk = k0 + bkr;
      bf16x8 q = {0, 0, 0, 0, 0, 0, 0, 0};
      if (gk < Kdim) {
        if (wp_v8)
          q = *(const bf16x8*)(wp + wp_base);
        else {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            if (n_blk + bc8 + e < sh.C)
              ((unsigned short*)&q)[e] = wp[wp_base + e];
        }
      }
      *(bf16x8*)rb = q;
    }
    wp_base += (long)BKB * sh.C;
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int j = 0; j < 2; ++j)
      *(bf16x8*)&A_lds[buf][(am + j * 64) * LDA_B + ak] = *(bf16x8*)ra[j];
#pragma unroll
    for (int e = 0; e < 8; ++e)
      B_lds[buf][(bc8 + e) * LDT_B + bkr] = rb[e];
  };

  stage_load(0);
  stage_write(0);
  if (BKB < Kdim) stage_load(BKB);
  __syncthreads();
  int buf = 0;
  for (int k0 = 0; k0 < Kdim; k0 += BKB) {
    if (k0 + BKB < Kdim) {
      stage_write(buf ^ 1);
      if (k0 + 2 * BKB < Kdim) stage_load(k0 + 2 * BKB);
    }
    const unsigned short* Ab = A_lds[buf];
    const unsigned short* Bb = B_lds[buf];
    bf16x8 a_frag[MI], b_frag[NI];
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
      a_frag[mi] = *(const bf16x8*)&Ab[(wr * 64 + mi * 16 + l15) * LDA_B +
                                       l4 * 8];
#pragma unroll
    for (int ni = 0; ni < NI; ++ni)
      b_frag[ni] = *(const bf16x8*)&Bb[(wc * 32 + ni * 16 + l15) * LDT_B +
                                       l4 * 8];
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
#pragma unroll
      for (int ni = 0; ni < NI; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    __syncthreads();
    buf ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      int c = n_blk + wc * 32 + ni * 16 + l15;
      if (c >= sh.C) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long m = m_blk + wr * 64 + mi * 16 + l4 * 4 + r;
        if (m >= M) continue;
        float v = acc[mi][ni][r];
        if (relu_y && bf2f_(relu_y[m * sh.C + c]) <= 0.f) v = 0.f;
        dx[m * sh.C + c] = f2bf_(v);
      }
    }
}

// ------------------------------------------------------------ bwd-weight

// BM=64(ko) x BNW(crs) x BK=32(m); A = dy^T, B = x-patch^T — both staged
// transposed for contiguous 8-element m fragments.  fp32 slabs/output
// (weight grads stay fp32); reuses the f32 split-K reduce + dw permute.
// BNW=64: 2x2 fragments (4 MFMA/wave/step, 19 KB LDS); BNW=128: 2x4
// fragments (8 MFMA/wave/step, 31 KB LDS, halves dy re-reads) — variants
// compared on-box by tests/perf/bwdw_micro.
template <bool P0, int BNW, int DEPTH>
__global__ __launch_bounds__(256)
void conv_bwd_weight_bf16_k(const unsigned short* __restrict__ dy,
                            const unsigned short* __restrict__ x,
                            float* __restrict__ out, ConvShapeB sh,
                            int Ncrs, long k_per_chunk, int direct_out) {
  constexpr int BM = 64, MI = 2, NI = BNW / 32;
  constexpr int TB = BNW / 8;      // threads per m-row of B staging
  constexpr int BR = TB / 8;       // B staging rounds (m rows per thread)
  __shared__ unsigned short A_lds[2][BM * LDA_B];   // [ko][m]
  __shared__ unsigned short B_lds[2][BNW * LDT_B];  // [crs][m]
  const int t = threadIdx.x;
  const int wave = t >> 6, lane = t & 63;
  const int wr = wave >> 1, wc = wave & 1;
  const int l15 = lane & 15, l4 = lane >> 4;
  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  const int m_blk = blockIdx.x * BM;   // over ko
  const int n_blk = blockIdx.y * BNW;  // over crs
  const long Kdim = (long)sh.Nb * sh.OH * sh.OW;
  const long k_lo = (long)blockIdx.z * k_per_chunk;
  const long k_hi = min(Kdim, k_lo + k_per_chunk);

  // A: thread loads dy[m][ko8..ko8+7]; ko8 = (t&7)*8, m = t>>3 (32 m)
  const int ako = (t & 7) * 8, amr = t >> 3;
  // B: thread loads x-patch[m][crs8..+7] (8 contiguous c) for BR m rows
  const int bcr = (t % TB) * 8, bmr = t / TB;

  // (r,s,c0) for this thread's B columns (fixed)
  int br_, bs_, bc0_;
  {
    int crs = min(n_blk + bcr, Ncrs - 1);
    int rs = crs / sh.C;
    br_ = rs / sh.S;
    bs_ = rs % sh.S;
    bc0_ = crs - rs * sh.C;
  }

  // DEPTH-1 register sets ring between global load and LDS write: the
  // global gather for k0 + DEPTH*BK is issued while k0 computes, giving
  // (DEPTH-1) k-steps of latency budget for the scattered x-patch loads
  unsigned short raA[DEPTH - 1][8], rbB[DEPTH - 1][BR][8];
  // B-side (n,oh,ow) coordinates are advanced INCREMENTALLY by BKB per
  // stage instead of div/mod per load: the k->(n,oh,ow) divisions cost
  // ~60 VALU cycles/step vs ~64 MFMA cycles/step — they were the
  // bottleneck (tests/perf/bwdw_micro: all layers plateaued ~170 us).
  int c_ow[BR], c_oh[BR];
  long c_nb[BR], c_k[BR];
  const int dow = BKB % sh.OW, doh = BKB / sh.OW;
#pragma unroll
  for (int j = 0; j < BR; ++j) {
    long k = k_lo + bmr + j * (256 / TB);
    c_k[j] = k;
    c_ow[j] = (int)(k % sh.OW);
    c_oh[j] = (int)((k / sh.OW) % sh.OH);
    c_nb[j] = k / ((long)sh.OW * sh.OH);
  }
  auto stage_load = [&](int set, long k0) {
    {
      long k = k0 + amr;
      bf16x8 q = {0, 0, 0, 0, 0, 0, 0, 0};
      if (k < k_hi) {
        if ((sh.Kout % 8) == 0 && m_blk + ako + 7 < sh.Kout)
          q = *(const bf16x8*)(dy + k * sh.Kout + m_blk + ako);
        else {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            if (m_blk + ako + e < sh.Kout)
              ((unsigned short*)&q)[e] = dy[k * sh.Kout + m_blk + ako + e];
        }
      }
      *(bf16x8*)raA[set] = q;
    }
#pragma unroll
    for (int j = 0; j < BR; ++j) {
      bf16x8 q = {0, 0, 0, 0, 0, 0, 0, 0};
      if (c_k[j] < k_hi) {
        int ih = c_oh[j] * sh.stride - sh.pad + br_;
        int iw = c_ow[j] * sh.stride - sh.pad + bs_;
        if (n_blk + bcr < Ncrs &&
            (P0 || (ih >= 0 && ih < sh.H && iw >= 0 && iw < sh.W)))
          q = *(const bf16x8*)(x + (c_nb[j] * sh.H * sh.W +
                                    (long)ih * sh.W + iw) * sh.C + bc0_);
      }
      *(bf16x8*)rbB[set][j] = q;
      // advance to this slot's next stage (k += BKB)
      c_k[j] += BKB;
      c_ow[j] += dow;
      if (c_ow[j] >= sh.OW) { c_ow[j] -= sh.OW; ++c_oh[j]; }
      c_oh[j] += doh;
      while (c_oh[j] >= sh.OH) { c_oh[j] -= sh.OH; ++c_nb[j]; }
    }
  };
  auto stage_write = [&](int buf, int set) {
    int m = amr;
#pragma unroll
    for (int e = 0; e < 8; ++e)
      A_lds[buf][(ako + e) * LDA_B + m] = raA[set][e];
#pragma unroll
    for (int j = 0; j < BR; ++j)
#pragma unroll
      for (int e = 0; e < 8; ++e)
        B_lds[buf][(bcr + e) * LDT_B + bmr + j * (256 / TB)] = rbB[set][j][e];
  };

  auto mfma_step = [&](int buf) {
    const unsigned short* Ab = A_lds[buf];
    const unsigned short* Bb = B_lds[buf];
    bf16x8 a_frag[MI], b_frag[NI];
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
      a_frag[mi] = *(const bf16x8*)&Ab[(wr * 32 + mi * 16 + l15) * LDA_B +
                                       l4 * 8];
#pragma unroll
    for (int ni = 0; ni < NI; ++ni)
      b_frag[ni] = *(const bf16x8*)&Bb[(wc * (BNW / 2) + ni * 16 + l15) *
                                           LDT_B +
                                       l4 * 8];
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
#pragma unroll
      for (int ni = 0; ni < NI; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
  };

  stage_load(0, k_lo);
  stage_write(0, 0);
  if (k_lo + BKB < k_hi) stage_load(0, k_lo + BKB);
  if (DEPTH > 2 && k_lo + 2 * BKB < k_hi) stage_load(1, k_lo + 2 * BKB);
  __syncthreads();
  int buf = 0;
  if (DEPTH > 2) {
    // 2x-unrolled so the register-ring set index is a LITERAL at every
    // call site — a runtime index forces the ring into scratch (measured
    // 2.5x regression, profiles/r01_bwdw_micro.md)
    long k0 = k_lo;
    while (k0 < k_hi) {
      if (k0 + BKB < k_hi) {
        stage_write(buf ^ 1, 0);
        if (k0 + 3 * BKB < k_hi) stage_load(0, k0 + 3 * BKB);
      }
      mfma_step(buf);
      __syncthreads();
      buf ^= 1;
      k0 += BKB;
      if (k0 >= k_hi) break;
      if (k0 + BKB < k_hi) {
        stage_write(buf ^ 1, 1);
        if (k0 + 3 * BKB < k_hi) stage_load(1, k0 + 3 * BKB);
      }
      mfma_step(buf);
      __syncthreads();
      buf ^= 1;
      k0 += BKB;
    }
  } else {
    for (long k0 = k_lo; k0 < k_hi; k0 += BKB) {
      if (k0 + BKB < k_hi) {
        stage_write(buf ^ 1, 0);
        if (k0 + 2 * BKB < k_hi) stage_load(0, k0 + 2 * BKB);
      }
      mfma_step(buf);
      __syncthreads();
      buf ^= 1;
    }
  }

#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      int crs = n_blk + wc * (BNW / 2) + ni * 16 + l15;
      if (crs >= Ncrs) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int ko = m_blk + wr * 32 + mi * 16 + l4 * 4 + r;
        if (ko >= sh.Kout) continue;
        if (direct_out)
          out[(long)ko * Ncrs + crs] = acc[mi][ni][r];
        else
          out[((long)blockIdx.z * sh.Kout + ko) * Ncrs + crs] =
              acc[mi][ni][r];
      }
    }
}

// w (fp32, (KO,C,R,S)) -> bf16 staged layouts
__global__ void wperm_rsc_ko_bf16_k(const float* __restrict__ w,
                                    unsigned short* __restrict__ out,
                                    int Kout, int C, int RS) {
  long n = (long)Kout * C * RS;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int rs = i % RS;
    int c = (i / RS) % C;
    long ko = i / ((long)RS * C);
    out[((long)rs * C + c) * Kout + ko] = f2bf_(w[i]);
  }
}

__global__ void wperm_rsko_c_bf16_k(const float* __restrict__ w,
                                    unsigned short* __restrict__ out,
                                    int Kout, int C, int RS) {
  long n = (long)Kout * C * RS;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int rs = i % RS;
    int c = (i / RS) % C;
    long ko = i / ((long)RS * C);
    out[((long)rs * Kout + ko) * C + c] = f2bf_(w[i]);
  }
}

// bf16 NHWC conv-bias gradient: db[ko] = column sums of dy [M][KO]
// (full-block slice mapping as in the f32 conv_db_stage1_k)
__global__ void conv_db_bf16_stage1_k(const unsigned short* __restrict__ dy,
                                      float* __restrict__ partials, long M,
                                      int Kout, int chunks, int K_blk) {
  int sub_per = blockDim.x / K_blk;
  int ko = blockIdx.y * K_blk + threadIdx.x % K_blk;
  int chunk = blockIdx.x * sub_per + threadIdx.x / K_blk;
  if (ko >= Kout || chunk >= chunks) return;
  long per = (M + chunks - 1) / chunks;
  long lo = (long)chunk * per, hi = min(M, lo + per);
  float acc = 0.f;
#pragma unroll 4
  for (long m = lo; m < hi; ++m) acc += bf2f_(dy[m * Kout + ko]);
  partials[(long)chunk * Kout + ko] = acc;
}

extern "C" {
void launch_splitk_reduce(const float* ws, float* C, const float* bias,
                          int M, int N, int ldc, int SK, int relu, void* s);
int conv_bwd_weight_splitk(int Kout, int Ncrs, long Kdim);

// bf16 split-K policy: target ~2304 blocks (9/CU queued) — the SK sweep
// on MI355X showed the f32 path's 1024-block target leaves 10-20% on the
// table for every ResNet layer shape (tests/perf/bwdw_micro)
int conv_bwd_weight_bf16_splitk(int Kout, int Ncrs, long Kdim) {
  long tiles = ((Kout + 63) / 64) * (long)((Ncrs + 63) / 64);
  if (tiles >= 2304 || Kdim <= 2 * BKB) return 1;
  long want = (2304 + tiles - 1) / tiles;
  long max_chunks = (Kdim + BKB - 1) / BKB;
  long sk = want < max_chunks ? want : max_chunks;
  return (int)(sk < 1 ? 1 : (sk > 256 ? 256 : sk));
}
void launch_dwperm_rsc_crs(const float*, float*, int, int, int, void*);
void launch_conv_db_stage2(const float*, float*, int, int, void*);
int conv_db_chunks(long M, int Kout);

void launch_conv_fwd_bf16(const unsigned short* x, const unsigned short* wt,
                          const float* bias, unsigned short* y, int Nb,
                          int C, int H, int W, int Kout, int R, int S,
                          int OH, int OW, int stride, int pad, int relu,
                          void* s) {
  ConvShapeB sh{Nb, C, H, W, Kout, R, S, OH, OW, stride, pad};
  int Kdim = C * R * S;
  long M = (long)Nb * OH * OW;
  dim3 grid((M + 127) / 128, (Kout + 63) / 64, 1);
  hipStream_t st = (hipStream_t)s;
  if (pad == 0)
    conv_fwd_bf16_k<true><<<grid, 256, 0, st>>>(x, wt, bias, y, sh, Kdim,
                                                relu);
  else
    conv_fwd_bf16_k<false><<<grid, 256, 0, st>>>(x, wt, bias, y, sh, Kdim,
                                                 relu);
}

void launch_conv_bwd_data_bf16_relu(const unsigned short* dy,
                                    const unsigned short* wp,
                                    unsigned short* dx,
                                    const unsigned short* relu_y, int Nb,
                                    int C, int H, int W, int Kout, int R,
                                    int S, int OH, int OW, int stride,
                                    int pad, void* s) {
  ConvShapeB sh{Nb, C, H, W, Kout, R, S, OH, OW, stride, pad};
  int Kdim = Kout * R * S;
  long M = (long)Nb * H * W;
  dim3 grid((M + 127) / 128, (C + 63) / 64, 1);
  hipStream_t st = (hipStream_t)s;
  if (stride == 1)
    conv_bwd_data_bf16_k<1><<<grid, 256, 0, st>>>(dy, wp, dx, sh, Kdim,
                                                  relu_y);
  else if (stride == 2)
    conv_bwd_data_bf16_k<2><<<grid, 256, 0, st>>>(dy, wp, dx, sh, Kdim,
                                                  relu_y);
  else
    conv_bwd_data_bf16_k<0><<<grid, 256, 0, st>>>(dy, wp, dx, sh, Kdim,
                                                  relu_y);
}

void launch_conv_bwd_data_bf16(const unsigned short* dy,
                               const unsigned short* wp, unsigned short* dx,
                               int Nb, int C, int H, int W, int Kout, int R,
                               int S, int OH, int OW, int stride, int pad,
                               void* s) {
  launch_conv_bwd_data_bf16_relu(dy, wp, dx, nullptr, Nb, C, H, W, Kout, R,
                                 S, OH, OW, stride, pad, s);
}

// variant 0: 64-wide tile depth-2; 1: 128-wide; 2: 64-wide depth-3 ring
void launch_conv_bwd_weight_bf16_ex(const unsigned short* dy,
                                    const unsigned short* x, float* dw,
                                    float* ws, int SK, int variant, int Nb,
                                    int C, int H, int W, int Kout, int R,
                                    int S, int OH, int OW, int stride,
                                    int pad, void* s) {
  ConvShapeB sh{Nb, C, H, W, Kout, R, S, OH, OW, stride, pad};
  int Ncrs = C * R * S;
  long Kdim = (long)Nb * OH * OW;
  long k_per_chunk =
      SK == 1 ? Kdim : (((Kdim + SK - 1) / SK + BKB - 1) / BKB) * BKB;
  int bnw = variant == 1 ? 128 : 64;
  dim3 grid((Kout + 63) / 64, (Ncrs + bnw - 1) / bnw, SK);
  hipStream_t st = (hipStream_t)s;
  float* slabs = ws;
  float* rsc = ws + (long)SK * Kout * Ncrs;
  float* target = SK == 1 ? rsc : slabs;
  if (variant == 1) {
    if (pad == 0)
      conv_bwd_weight_bf16_k<true, 128, 2><<<grid, 256, 0, st>>>(
          dy, x, target, sh, Ncrs, k_per_chunk, SK == 1);
    else
      conv_bwd_weight_bf16_k<false, 128, 2><<<grid, 256, 0, st>>>(
          dy, x, target, sh, Ncrs, k_per_chunk, SK == 1);
  } else if (variant == 2) {
    if (pad == 0)
      conv_bwd_weight_bf16_k<true, 64, 3><<<grid, 256, 0, st>>>(
          dy, x, target, sh, Ncrs, k_per_chunk, SK == 1);
    else
      conv_bwd_weight_bf16_k<false, 64, 3><<<grid, 256, 0, st>>>(
          dy, x, target, sh, Ncrs, k_per_chunk, SK == 1);
  } else {
    if (pad == 0)
      conv_bwd_weight_bf16_k<true, 64, 2><<<grid, 256, 0, st>>>(
          dy, x, target, sh, Ncrs, k_per_chunk, SK == 1);
    else
      conv_bwd_weight_bf16_k<false, 64, 2><<<grid, 256, 0, st>>>(
          dy, x, target, sh, Ncrs, k_per_chunk, SK == 1);
  }
  extern void launch_splitk_reduce_dwperm(const float*, float*, int, int,
                                          int, int, void*);
  launch_splitk_reduce_dwperm(SK == 1 ? rsc : slabs, dw, Kout, C, R * S, SK,
                              s);
}

void launch_conv_bwd_weight_bf16(const unsigned short* dy,
                                 const unsigned short* x, float* dw,
                                 float* ws, int SK, int Nb, int C, int H,
                                 int W, int Kout, int R, int S, int OH,
                                 int OW, int stride, int pad, void* s) {
  // depth-3 ring (variant 2) wins 3.5-4% on every 3x3 shape at the policy
  // SK and ties on 1x1 (bwdw_micro run 4) — production default
  launch_conv_bwd_weight_bf16_ex(dy, x, dw, ws, SK, 2, Nb, C, H, W, Kout,
                                 R, S, OH, OW, stride, pad, s);
}

void launch_wperm_rsc_ko_bf16(const float* w, unsigned short* out, int Kout,
                              int C, int RS, void* s) {
  wperm_rsc_ko_bf16_k<<<grid_for((long)Kout * C * RS), kBlock, 0,
                        (hipStream_t)s>>>(w, out, Kout, C, RS);
}
void launch_wperm_rsko_c_bf16(const float* w, unsigned short* out, int Kout,
                              int C, int RS, void* s) {
  wperm_rsko_c_bf16_k<<<grid_for((long)Kout * C * RS), kBlock, 0,
                        (hipStream_t)s>>>(w, out, Kout, C, RS);
}

void launch_conv_db_bf16(const unsigned short* dy, float* db,
                         float* partials, int Nb, int Kout, int OHW,
                         void* s) {
  hipStream_t st = (hipStream_t)s;
  long M = (long)Nb * OHW;
  int chunks = conv_db_chunks(M, Kout);
  int K_blk = Kout >= kBlock ? kBlock
                             : ((kBlock % Kout == 0) ? Kout : kBlock);
  int sub_per = kBlock / K_blk;
  dim3 g1((chunks + sub_per - 1) / sub_per, (Kout + K_blk - 1) / K_blk);
  conv_db_bf16_stage1_k<<<g1, kBlock, 0, st>>>(dy, partials, M, Kout,
                                               chunks, K_blk);
  launch_conv_db_stage2(partials, db, Kout, chunks, s);
}
}
