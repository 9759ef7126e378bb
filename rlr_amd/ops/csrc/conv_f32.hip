// Implicit-GEMM fp32 convolution on MFMA — NHWC.
//
// Activations are channels_last (N,H,W,C storage): the K-dimension of the
// implicit GEMM is ordered so its fastest index is the CONTIGUOUS channel
// axis, which turns the patch gathers into float4 loads with ONE address
// computation per 4 elements and makes the output stores coalesced.  The
// first NCHW version of these kernels measured 13.7 VALU instructions per
// MFMA (profiles/r01_bench_kernel_stats.md) — the gather address math, not
// the matrix pipe, was the limiter.
//
//   fwd:        M=Nb*OH*OW, N=K_out, Kdim=(r,s,c) — C%32==0 makes a 32-wide
//               K-tile sit inside one (r,s) tap: the tap decomposition is
//               computed once per tile, not per element
//   bwd-data:   M=Nb*H*W, N=C, Kdim=(r,s,ko), KO%32==0 fast path
//   bwd-weight: M=K_out (64x64 tile), N=(r,s,c), Kdim=Nb*OH*OW, split-K
//               slabs + fixed-order reduce; output permuted (r,s,c)->(c,r,s)
//               to match the torch (KO,C,R,S) weight layout
//
// Weight transforms (tiny, per call): fwd wt[(r,s,c)][ko], bwd-data
// wp[(r,s,ko)][c].  All reductions fixed-order; no atomics.
#include "common.h"

typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr int BK = 32;
constexpr int LDA_S = BK + 2;
constexpr int BN = 64;
// +8 pad: consecutive k rows land 8 banks apart (mild 2-way conflict on
// half of each b32 lane group) but the 128x64x32 tiles' LDS drops to
// 53.2 KB -> 3 blocks/CU instead of 2 (more TLP to hide staging).
constexpr int LDB_S = BN + 8;

struct ConvShape {
  int Nb, C, H, W, Kout, R, S, OH, OW, stride, pad;
};

// Fragment compute over one staged K-tile; wave grid 2x2, each wave owns
// an (MI*16 x NI*16) subtile.
template <int MI, int NI, int LDA = LDA_S, int LDB = LDB_S>
__device__ __forceinline__ void mfma_tile(const float* __restrict__ Abuf,
                                          const float* __restrict__ Bbuf,
                                          int wr, int wc, int l15, int l4,
                                          f32x4 (&acc)[MI][NI]) {
#pragma unroll
  for (int kk = 0; kk < BK / 4; ++kk) {
    float a_frag[MI], b_frag[NI];
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
      a_frag[mi] =
          Abuf[(wr * MI * 16 + mi * 16 + l15) * LDA + kk * 4 + l4];
#pragma unroll
    for (int ni = 0; ni < NI; ++ni)
      b_frag[ni] =
          Bbuf[(kk * 4 + l4) * LDB + wc * NI * 16 + ni * 16 + l15];
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
#pragma unroll
      for (int ni = 0; ni < NI; ++ni)
        acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x4f32(
            a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
  }
}

// ------------------------------------------------------------------- fwd

// V4: C % 32 == 0 (vectorized channel gather).  P0: pad == 0 (no bounds).
template <bool P0, bool V4>
__global__ __launch_bounds__(256)
void conv_fwd_k(const float* __restrict__ x, const float* __restrict__ wt,
                const float* __restrict__ bias, float* __restrict__ y,
                ConvShape sh, int Kdim, int relu) {
  constexpr int BM = 128, MI = 4, NI = 2;
  __shared__ float A_lds[2][BM * LDA_S];
  __shared__ float B_lds[2][BK * LDB_S];
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
  const int am = t >> 3, ak = (t & 7) * 4;
  const int bk = t >> 4, bn = (t & 15) * 4;

  // per-thread m -> (nb, oh, ow), once
  int ow0[BM / 32], oh0[BM / 32];
  long base[BM / 32];  // V4: patch base incl. tap 0 + ak; else nb*H*W*C
  bool mval[BM / 32];
#pragma unroll
  for (int j = 0; j < BM / 32; ++j) {
    long gm = m_blk + am + j * 32;
    mval[j] = gm < M;
    long gmc = mval[j] ? gm : 0;
    int ow = gmc % sh.OW;
    int oh = (gmc / sh.OW) % sh.OH;
    long nb = gmc / ((long)sh.OW * sh.OH);
    ow0[j] = ow * sh.stride - sh.pad;
    oh0[j] = oh * sh.stride - sh.pad;
    base[j] = nb * (long)sh.H * sh.W * sh.C;
    if (V4)  // fold the per-thread patch origin into the base: the gather
             // address becomes base[j] + tap_off (ONE add per load)
      base[j] += ((long)oh0[j] * sh.W + ow0[j]) * sh.C + ak;
  }

  // V4 staged-tap state: stage_load is called with strictly ascending k0
  // (0, BK, 2BK, ...), so the (r,s,c)->offset decomposition is carried
  // incrementally — no div/mod in the loop.  tap_off = (r*W+s)*C + cb.
  int tr = 0, ts = 0, tcb = 0;
  long tap_off = 0;
  const bool wt_v4 = (sh.Kout % 4) == 0 && n_blk + bn + 3 < sh.Kout;
  const bool wt_any = n_blk + bn < sh.Kout;
  long wt_base = (long)bk * sh.Kout + n_blk + bn;  // advances BK*Kout/call
  const long wt_j16 = 16L * sh.Kout;

  float ra[BM / 32][4];
  float4 rb[2];
  auto stage_load = [&](int k0) {
    if (V4) {
      const int r = tr, s = ts;
      const long toff = tap_off;
#pragma unroll
      for (int j = 0; j < BM / 32; ++j) {
        float4 q = {0.f, 0.f, 0.f, 0.f};
        bool ok = mval[j];
        if (!P0) {
          int ih = oh0[j] + r, iw = ow0[j] + s;
          ok = ok && (unsigned)ih < (unsigned)sh.H &&
               (unsigned)iw < (unsigned)sh.W;
        }
        if (ok) q = *(const float4*)(x + base[j] + toff);
        ra[j][0] = q.x; ra[j][1] = q.y; ra[j][2] = q.z; ra[j][3] = q.w;
      }
      // advance tap state to the next K-tile (C % 32 == 0 in V4: a c-
      // rollover lands exactly on tile boundaries)
      tcb += BK; tap_off += BK;
      if (tcb == sh.C) {
        tcb = 0;
        if (++ts == sh.S) { ts = 0; ++tr;
                            tap_off += (long)(sh.W - sh.S) * sh.C; }
      }
    } else {
#pragma unroll
      for (int j = 0; j < BM / 32; ++j) {
        ra[j][0] = ra[j][1] = ra[j][2] = ra[j][3] = 0.f;
        if (mval[j]) {
#pragma unroll
          for (int e = 0; e < 4; ++e) {
            int k = k0 + ak + e;
            if (k < Kdim) {
              int c = k % sh.C;
              int rs = k / sh.C;
              int r = rs / sh.S, s = rs % sh.S;
              int ih = oh0[j] + r, iw = ow0[j] + s;
              if (P0 || (ih >= 0 && ih < sh.H && iw >= 0 && iw < sh.W))
                ra[j][e] =
                    x[base[j] + ((long)ih * sh.W + iw) * sh.C + c];
            }
          }
        }
      }
    }
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int gk = k0 + bk + j * 16;
      float4 q = {0.f, 0.f, 0.f, 0.f};
      if (gk < Kdim) {
        const float* src = wt + wt_base + j * wt_j16;
        if (wt_v4)
          q = *(const float4*)src;
        else if (wt_any) {
          if (n_blk + bn + 0 < sh.Kout) q.x = src[0];
          if (n_blk + bn + 1 < sh.Kout) q.y = src[1];
          if (n_blk + bn + 2 < sh.Kout) q.z = src[2];
          if (n_blk + bn + 3 < sh.Kout) q.w = src[3];
        }
      }
      rb[j] = q;
    }
    wt_base += (long)BK * sh.Kout;
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int j = 0; j < BM / 32; ++j) {
      float* dst = &A_lds[buf][(am + j * 32) * LDA_S + ak];
      ((float2*)dst)[0] = {ra[j][0], ra[j][1]};
      ((float2*)dst)[1] = {ra[j][2], ra[j][3]};
    }
#pragma unroll
    for (int j = 0; j < 2; ++j)
      *(float4*)&B_lds[buf][(bk + j * 16) * LDB_S + bn] = rb[j];
  };

  stage_load(0);
  stage_write(0);
  if (BK < Kdim) stage_load(BK);
  __syncthreads();
  int buf = 0;
  for (int k0 = 0; k0 < Kdim; k0 += BK) {
    if (k0 + BK < Kdim) {
      stage_write(buf ^ 1);
      if (k0 + 2 * BK < Kdim) stage_load(k0 + 2 * BK);
    }
    mfma_tile<MI, NI>(A_lds[buf], B_lds[buf], wr, wc, l15, l4, acc);
    __syncthreads();
    buf ^= 1;
  }

  // NHWC epilogue: ko = col is the contiguous axis -> coalesced stores
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
        y[m * sh.Kout + ko] = v;
      }
    }
}

// -------------------------------------------- fwd, x-resident variant
// For stride-1 pad-0 C%32==0 layers whose image has >= 128 output pixels
// (conv2-class): the block's whole input region is loaded into LDS ONCE
// (instead of re-gathering each patch element per K-tile — a 9x re-read
// through L3) and A fragments are read directly with a per-pixel +1 bank
// pad.  One image per block; B (weights) stays double-buffer streamed.
__global__ __launch_bounds__(256)
void conv_fwd_res_k(const float* __restrict__ x,
                    const float* __restrict__ wt,  // [(r,s,c)][KO]
                    const float* __restrict__ bias, float* __restrict__ y,
                    ConvShape sh, int Kdim, int relu, int ptiles,
                    int lds_rows) {
  constexpr int BM = 128, MI = 4, NI = 2;
  extern __shared__ __attribute__((aligned(16))) float dynLds[];
  float* x_lds = dynLds;                       // [lds_rows][W][C+1-padded]
  // B buffers start 16-B aligned after the (odd-strided) x region
  long xf = ((long)lds_rows * sh.W * (sh.C + 1) + 3) & ~3L;
  float* B_lds0 = dynLds + xf;
  // B double buffer after the x region
  const int t = threadIdx.x;
  const int wave = t >> 6, lane = t & 63;
  const int wr = wave >> 1, wc = wave & 1;
  const int l15 = lane & 15, l4 = lane >> 4;
  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  const int nb = blockIdx.x / ptiles;
  const int p0 = (blockIdx.x % ptiles) * BM;
  const int OHOW = sh.OH * sh.OW;
  const int n_blk = blockIdx.y * BN;
  const int Cp = sh.C + 1;

  const int oh_lo = p0 / sh.OW;
  // ---- load the x region once (always in-bounds: pad==0, stride==1) ----
  {
    const float* xsrc =
        x + ((long)nb * sh.H + oh_lo) * sh.W * sh.C;
    int total4 = lds_rows * sh.W * sh.C / 4;
    int avail_rows = sh.H - oh_lo;
    int total4_avail = min(total4, avail_rows * sh.W * sh.C / 4);
    for (int i = t; i < total4_avail; i += 256) {
      float4 q = *(const float4*)(xsrc + (long)i * 4);
      int c = (i * 4) % sh.C;
      int pix = (i * 4) / sh.C;
      float* dst = &x_lds[pix * Cp + c];
      dst[0] = q.x; dst[1] = q.y; dst[2] = q.z; dst[3] = q.w;
    }
  }

  // per-lane fragment bases: lds pixel offset of each mi's output pixel
  int pixrel[MI];
  bool mval[MI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
    int m = p0 + wr * 64 + mi * 16 + l15;
    mval[mi] = m < OHOW;
    int mm = mval[mi] ? m : 0;
    int oh = mm / sh.OW, ow = mm % sh.OW;
    pixrel[mi] = ((oh - oh_lo) * sh.W + ow) * Cp;
  }

  // ---- B staging (T14 split, double-buffered) ----
  const int bk = t >> 4, bn = (t & 15) * 4;
  const bool wt_v4 = (sh.Kout % 4) == 0 && n_blk + bn + 3 < sh.Kout;
  const bool wt_any = n_blk + bn < sh.Kout;
  long wt_base = (long)bk * sh.Kout + n_blk + bn;
  const long wt_j16 = 16L * sh.Kout;
  float4 rb[2];
  auto stage_loadB = [&](int k0) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int gk = k0 + bk + j * 16;
      float4 q = {0.f, 0.f, 0.f, 0.f};
      if (gk < Kdim) {
        const float* srcp = wt + wt_base + j * wt_j16;
        if (wt_v4)
          q = *(const float4*)srcp;
        else if (wt_any) {
          if (n_blk + bn + 0 < sh.Kout) q.x = srcp[0];
          if (n_blk + bn + 1 < sh.Kout) q.y = srcp[1];
          if (n_blk + bn + 2 < sh.Kout) q.z = srcp[2];
          if (n_blk + bn + 3 < sh.Kout) q.w = srcp[3];
        }
      }
      rb[j] = q;
    }
    wt_base += (long)BK * sh.Kout;
  };
  auto stage_writeB = [&](int buf) {
#pragma unroll
    for (int j = 0; j < 2; ++j)
      *(float4*)&B_lds0[buf * BK * LDB_S + (bk + j * 16) * LDB_S + bn] =
          rb[j];
  };

  stage_loadB(0);
  stage_writeB(0);
  if (BK < Kdim) stage_loadB(BK);
  __syncthreads();
  int buf = 0;
  // incremental tap state (one tap per 32-wide K-tile; C % 32 == 0 here):
  // tap = (r*W + s)*Cp + c0 + l4, carried across iterations — no div/mod
  int ts = 0, tcb = 0, tap = l4;
  for (int k0 = 0; k0 < Kdim; k0 += BK) {
    if (k0 + BK < Kdim) {
      stage_writeB(buf ^ 1);
      if (k0 + 2 * BK < Kdim) stage_loadB(k0 + 2 * BK);
    }
    const float* Bbuf = &B_lds0[buf * BK * LDB_S];
#pragma unroll
    for (int kk = 0; kk < BK / 4; ++kk) {
      float a_frag[MI], b_frag[NI];
#pragma unroll
      for (int mi = 0; mi < MI; ++mi)
        a_frag[mi] = x_lds[pixrel[mi] + tap + kk * 4];
#pragma unroll
      for (int ni = 0; ni < NI; ++ni)
        b_frag[ni] =
            Bbuf[(kk * 4 + l4) * LDB_S + wc * 32 + ni * 16 + l15];
#pragma unroll
      for (int mi = 0; mi < MI; ++mi)
#pragma unroll
        for (int ni = 0; ni < NI; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              a_frag[mi], b_frag[ni], acc[mi][ni], 0, 0, 0);
    }
    tcb += BK; tap += BK;
    if (tcb == sh.C) {
      tcb = 0; tap += Cp - sh.C;  // +1: next s column in the padded row
      if (++ts == sh.S) { ts = 0; tap += (sh.W - sh.S) * Cp; }
    }
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
      for (int r2 = 0; r2 < 4; ++r2) {
        int m = p0 + wr * 64 + mi * 16 + l4 * 4 + r2;
        if (m >= OHOW) continue;
        float v = acc[mi][ni][r2];
        if (bias) v += bias[ko];
        if (relu) v = fmaxf(v, 0.f);
        y[((long)nb * OHOW + m) * sh.Kout + ko] = v;
      }
    }
}

// -------------------------------------------------------------- bwd-data

// Kdim order (r,s,ko); V4: KO % 32 == 0.  ST: compile-time stride.
template <int ST, bool V4>
__global__ __launch_bounds__(256)
void conv_bwd_data_k(const float* __restrict__ dy,
                     const float* __restrict__ wp,  // [(r,s,ko)][C]
                     float* __restrict__ dx, ConvShape sh, int Kdim,
                     const float* __restrict__ relu_y) {
  constexpr int BM = 128, MI = 4, NI = 2;
  __shared__ float A_lds[2][BM * LDA_S];
  __shared__ float B_lds[2][BK * LDB_S];
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
  const int am = t >> 3, ak = (t & 7) * 4;
  const int bk = t >> 4, bn = (t & 15) * 4;
  const int stride = ST > 0 ? ST : sh.stride;

  int iwp[BM / 32], ihp[BM / 32];
  long base[BM / 32];  // ST==1&&V4: folded patch base; else nb*OH*OW*KO
  bool mval[BM / 32];
#pragma unroll
  for (int j = 0; j < BM / 32; ++j) {
    long gm = m_blk + am + j * 32;
    mval[j] = gm < M;
    long gmc = mval[j] ? gm : 0;
    iwp[j] = (int)(gmc % sh.W) + sh.pad;
    ihp[j] = (int)((gmc / sh.W) % sh.H) + sh.pad;
    base[j] = (gmc / ((long)sh.W * sh.H)) * (long)sh.OH * sh.OW * sh.Kout;
    if (ST == 1 && V4)  // gather address = base[j] - tile_off (one sub)
      base[j] += ((long)ihp[j] * sh.OW + iwp[j]) * sh.Kout + ak;
  }

  // staged-tap state, advanced per stage_load call (ascending k0; KO%32==0
  // in V4 so ko-rollover lands on tile boundaries): (r,s) of the tap plus,
  // for ST==1, the folded offset noff = (r*OW+s)*KO - kb.
  int tr = 0, ts = 0, tkb = 0;
  long noff = 0;
  const bool wp_v4 = (sh.C % 4) == 0 && n_blk + bn + 3 < sh.C;
  const bool wp_any = n_blk + bn < sh.C;
  long wp_base = (long)bk * sh.C + n_blk + bn;
  const long wp_j16 = 16L * sh.C;

  float ra[BM / 32][4];
  float4 rb[2];
  auto stage_load = [&](int k0) {
    if (V4) {
      const int r = tr, s = ts;
      if (ST == 1) {
        const long toff = noff;
#pragma unroll
        for (int j = 0; j < BM / 32; ++j) {
          int ohn = ihp[j] - r, own = iwp[j] - s;
          float4 q = {0.f, 0.f, 0.f, 0.f};
          if (mval[j] && (unsigned)ohn < (unsigned)sh.OH &&
              (unsigned)own < (unsigned)sh.OW)
            q = *(const float4*)(dy + base[j] - toff);
          ra[j][0] = q.x; ra[j][1] = q.y; ra[j][2] = q.z; ra[j][3] = q.w;
        }
      } else {
        const int ko0 = tkb + ak;
#pragma unroll
        for (int j = 0; j < BM / 32; ++j) {
          int ohn = ihp[j] - r, own = iwp[j] - s;
          float4 q = {0.f, 0.f, 0.f, 0.f};
          if (mval[j] && ohn >= 0 && own >= 0 && ohn % stride == 0 &&
              own % stride == 0) {
            int oh = ohn / stride, ow = own / stride;
            if (oh < sh.OH && ow < sh.OW)
              q = *(const float4*)(dy + base[j] +
                                   ((long)oh * sh.OW + ow) * sh.Kout + ko0);
          }
          ra[j][0] = q.x; ra[j][1] = q.y; ra[j][2] = q.z; ra[j][3] = q.w;
        }
      }
      tkb += BK; noff -= BK;
      if (tkb == sh.Kout) {
        tkb = 0; noff += 2L * sh.Kout;
        if (++ts == sh.S) { ts = 0; ++tr;
                            noff += (long)(sh.OW - sh.S) * sh.Kout; }
      }
    } else {
#pragma unroll
      for (int j = 0; j < BM / 32; ++j) {
        ra[j][0] = ra[j][1] = ra[j][2] = ra[j][3] = 0.f;
        if (mval[j]) {
#pragma unroll
          for (int e = 0; e < 4; ++e) {
            int k = k0 + ak + e;
            if (k < Kdim) {
              int ko = k % sh.Kout;
              int rs = k / sh.Kout;
              int r = rs / sh.S, s = rs % sh.S;
              int ohn = ihp[j] - r, own = iwp[j] - s;
              if (ohn >= 0 && own >= 0 && ohn % stride == 0 &&
                  own % stride == 0) {
                int oh = ohn / stride, ow = own / stride;
                if (oh < sh.OH && ow < sh.OW)
                  ra[j][e] = dy[base[j] +
                                ((long)oh * sh.OW + ow) * sh.Kout + ko];
              }
            }
          }
        }
      }
    }
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int gk = k0 + bk + j * 16;
      float4 q = {0.f, 0.f, 0.f, 0.f};
      if (gk < Kdim) {
        const float* src = wp + wp_base + j * wp_j16;
        if (wp_v4)
          q = *(const float4*)src;
        else if (wp_any) {
          if (n_blk + bn + 0 < sh.C) q.x = src[0];
          if (n_blk + bn + 1 < sh.C) q.y = src[1];
          if (n_blk + bn + 2 < sh.C) q.z = src[2];
          if (n_blk + bn + 3 < sh.C) q.w = src[3];
        }
      }
      rb[j] = q;
    }
    wp_base += (long)BK * sh.C;
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int j = 0; j < BM / 32; ++j) {
      float* dst = &A_lds[buf][(am + j * 32) * LDA_S + ak];
      ((float2*)dst)[0] = {ra[j][0], ra[j][1]};
      ((float2*)dst)[1] = {ra[j][2], ra[j][3]};
    }
#pragma unroll
    for (int j = 0; j < 2; ++j)
      *(float4*)&B_lds[buf][(bk + j * 16) * LDB_S + bn] = rb[j];
  };

  stage_load(0);
  stage_write(0);
  if (BK < Kdim) stage_load(BK);
  __syncthreads();
  int buf = 0;
  for (int k0 = 0; k0 < Kdim; k0 += BK) {
    if (k0 + BK < Kdim) {
      stage_write(buf ^ 1);
      if (k0 + 2 * BK < Kdim) stage_load(k0 + 2 * BK);
    }
    mfma_tile<MI, NI>(A_lds[buf], B_lds[buf], wr, wc, l15, l4, acc);
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
        if (relu_y && relu_y[m * sh.C + c] <= 0.f) v = 0.f;
        dx[m * sh.C + c] = v;
      }
    }
}

// bwd-data, BN=32 (C <= 32): wave grid 4x1 (MI=2, NI=2 over 32 cols) —
// halves the MFMA work vs padding C=32 into a 64-wide tile.
constexpr int LDB32_S = 48;  // 32 + 16: consecutive k rows 16 banks apart

template <int ST, bool V4>
__global__ __launch_bounds__(256)
void conv_bwd_data32_k(const float* __restrict__ dy,
                       const float* __restrict__ wp,  // [(r,s,ko)][C]
                       float* __restrict__ dx, ConvShape sh, int Kdim,
                       const float* __restrict__ relu_y) {
  constexpr int BM = 128, MI = 2, NI = 2, BN32 = 32;
  __shared__ float A_lds[2][BM * LDA_S];
  __shared__ float B_lds[2][BK * LDB32_S];
  const int t = threadIdx.x;
  const int wave = t >> 6, lane = t & 63;
  const int wr = wave, wc = 0;  // 4x1 wave grid
  const int l15 = lane & 15, l4 = lane >> 4;
  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  const int m_blk = blockIdx.x * BM;
  const int n_blk = blockIdx.y * BN32;
  const long M = (long)sh.Nb * sh.H * sh.W;
  const int am = t >> 3, ak = (t & 7) * 4;
  const int bk = t >> 3, bn = (t & 7) * 4;  // 32x32 tile, 1 round
  const int stride = ST > 0 ? ST : sh.stride;

  int iwp[BM / 32], ihp[BM / 32];
  long base[BM / 32];  // ST==1&&V4: folded patch base (see bwd_data_k)
  bool mval[BM / 32];
#pragma unroll
  for (int j = 0; j < BM / 32; ++j) {
    long gm = m_blk + am + j * 32;
    mval[j] = gm < M;
    long gmc = mval[j] ? gm : 0;
    iwp[j] = (int)(gmc % sh.W) + sh.pad;
    ihp[j] = (int)((gmc / sh.W) % sh.H) + sh.pad;
    base[j] = (gmc / ((long)sh.W * sh.H)) * (long)sh.OH * sh.OW * sh.Kout;
    if (ST == 1 && V4)
      base[j] += ((long)ihp[j] * sh.OW + iwp[j]) * sh.Kout + ak;
  }

  int tr = 0, ts = 0, tkb = 0;
  long noff = 0;

  float ra[BM / 32][4];
  float4 rb;
  auto stage_load = [&](int k0) {
    if (V4) {
      const int r = tr, s = ts;
      if (ST == 1) {
        const long toff = noff;
#pragma unroll
        for (int j = 0; j < BM / 32; ++j) {
          int ohn = ihp[j] - r, own = iwp[j] - s;
          float4 q = {0.f, 0.f, 0.f, 0.f};
          if (mval[j] && (unsigned)ohn < (unsigned)sh.OH &&
              (unsigned)own < (unsigned)sh.OW)
            q = *(const float4*)(dy + base[j] - toff);
          ra[j][0] = q.x; ra[j][1] = q.y; ra[j][2] = q.z; ra[j][3] = q.w;
        }
      } else {
        const int ko0 = tkb + ak;
#pragma unroll
        for (int j = 0; j < BM / 32; ++j) {
          int ohn = ihp[j] - r, own = iwp[j] - s;
          float4 q = {0.f, 0.f, 0.f, 0.f};
          if (mval[j] && ohn >= 0 && own >= 0 && ohn % stride == 0 &&
              own % stride == 0) {
            int oh = ohn / stride, ow = own / stride;
            if (oh < sh.OH && ow < sh.OW)
              q = *(const float4*)(dy + base[j] +
                                   ((long)oh * sh.OW + ow) * sh.Kout + ko0);
          }
          ra[j][0] = q.x; ra[j][1] = q.y; ra[j][2] = q.z; ra[j][3] = q.w;
        }
      }
      tkb += BK; noff -= BK;
      if (tkb == sh.Kout) {
        tkb = 0; noff += 2L * sh.Kout;
        if (++ts == sh.S) { ts = 0; ++tr;
                            noff += (long)(sh.OW - sh.S) * sh.Kout; }
      }
    } else {
#pragma unroll
      for (int j = 0; j < BM / 32; ++j) {
        ra[j][0] = ra[j][1] = ra[j][2] = ra[j][3] = 0.f;
        if (mval[j]) {
#pragma unroll
          for (int e = 0; e < 4; ++e) {
            int k = k0 + ak + e;
            if (k < Kdim) {
              int ko = k % sh.Kout;
              int rs = k / sh.Kout;
              int r = rs / sh.S, s = rs % sh.S;
              int ohn = ihp[j] - r, own = iwp[j] - s;
              if (ohn >= 0 && own >= 0 && ohn % stride == 0 &&
                  own % stride == 0) {
                int oh = ohn / stride, ow = own / stride;
                if (oh < sh.OH && ow < sh.OW)
                  ra[j][e] = dy[base[j] +
                                ((long)oh * sh.OW + ow) * sh.Kout + ko];
              }
            }
          }
        }
      }
    }
    {
      int gk = k0 + bk;
      float4 q = {0.f, 0.f, 0.f, 0.f};
      if (gk < Kdim) {
        const float* src = wp + (long)gk * sh.C + n_blk + bn;
        if ((sh.C % 4) == 0 && n_blk + bn + 3 < sh.C)
          q = *(const float4*)src;
        else if (n_blk + bn < sh.C) {
          if (n_blk + bn + 0 < sh.C) q.x = src[0];
          if (n_blk + bn + 1 < sh.C) q.y = src[1];
          if (n_blk + bn + 2 < sh.C) q.z = src[2];
          if (n_blk + bn + 3 < sh.C) q.w = src[3];
        }
      }
      rb = q;
    }
  };
  auto stage_write = [&](int buf) {
#pragma unroll
    for (int j = 0; j < BM / 32; ++j) {
      float* dst = &A_lds[buf][(am + j * 32) * LDA_S + ak];
      ((float2*)dst)[0] = {ra[j][0], ra[j][1]};
      ((float2*)dst)[1] = {ra[j][2], ra[j][3]};
    }
    *(float4*)&B_lds[buf][bk * LDB32_S + bn] = rb;
  };

  stage_load(0);
  stage_write(0);
  if (BK < Kdim) stage_load(BK);
  __syncthreads();
  int buf = 0;
  for (int k0 = 0; k0 < Kdim; k0 += BK) {
    if (k0 + BK < Kdim) {
      stage_write(buf ^ 1);
      if (k0 + 2 * BK < Kdim) stage_load(k0 + 2 * BK);
    }
    mfma_tile<MI, NI, LDA_S, LDB32_S>(A_lds[buf], B_lds[buf], wr, wc, l15,
                                      l4, acc);
    __syncthreads();
    buf ^= 1;
  }

#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      int c = n_blk + ni * 16 + l15;
      if (c >= sh.C) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        long m = m_blk + wr * 32 + mi * 16 + l4 * 4 + r;
        if (m >= M) continue;
        float v = acc[mi][ni][r];
        if (relu_y && relu_y[m * sh.C + c] <= 0.f) v = 0.f;
        dx[m * sh.C + c] = v;
      }
    }
}

// ------------------------------------------------------------ bwd-weight

// 64x64 tile over (ko, crs=(r,s,c)); Kdim = m = (nb,oh,ow); split-K.
// A = dy^T gathered NHWC (float4 over ko, transposed into LDS);
// B = x patches (float4 over c when C%4==0).
// Slabs/output are in (r,s,c) column order; dwperm converts to (c,r,s).
template <bool P0, bool V4>
__global__ __launch_bounds__(256)
void conv_bwd_weight_k(const float* __restrict__ dy,
                       const float* __restrict__ x, float* __restrict__ out,
                       ConvShape sh, int Ncrs, long k_per_chunk,
                       int direct_out) {
  constexpr int BM = 64, MI = 2, NI = 2;
  constexpr int LDAW = 33;  // odd stride: transposed scalar writes 2-way
                            // instead of 4-way bank conflicted
  __shared__ float A_lds[2][BM * LDAW];
  __shared__ float B_lds[2][BK * LDB_S];
  const int t = threadIdx.x;
  const int wave = t >> 6, lane = t & 63;
  const int wr = wave >> 1, wc = wave & 1;
  const int l15 = lane & 15, l4 = lane >> 4;
  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = {0.f, 0.f, 0.f, 0.f};

  const int m_blk = blockIdx.x * BM;   // over Kout
  const int n_blk = blockIdx.y * BN;   // over (r,s,c)
  const long Kdim = (long)sh.Nb * sh.OH * sh.OW;
  const long k_lo = (long)blockIdx.z * k_per_chunk;
  const long k_hi = min(Kdim, k_lo + k_per_chunk);

  // A staging: float4 over ko (lanes 0-15 cover 64 consecutive ko); m from
  // the upper thread bits, 2 rounds cover BK=32 m.
  const int ako = (t & 15) * 4;
  const int amr = t >> 4;  // 0..15, +16 per round
  // B staging: float4 over crs (c fastest); 2 rounds over m.
  const int bn4 = (t & 15) * 4;
  const int bmr = t >> 4;

  // (r,s,c0) of this thread's B quad (fixed across tiles)
  int br_, bs_, bc0_;
  {
    int crs = min(n_blk + bn4, Ncrs - 1);
    int rs = crs / sh.C;
    br_ = rs / sh.S;
    bs_ = rs % sh.S;
    bc0_ = crs - rs * sh.C;
  }

  // hoisted thread-constant bounds + A-row base (advances BK*Kout/call)
  const bool a_v4 = (sh.Kout % 4) == 0 && m_blk + ako + 3 < sh.Kout;
  const bool b_v4q = n_blk + bn4 + 3 < Ncrs;
  const bool b_any = n_blk + bn4 < Ncrs;
  long a_base = (k_lo + amr) * (long)sh.Kout + m_blk + ako;
  const long a_j16 = 16L * sh.Kout;

  // per-j B-row pixel state, carried incrementally (+BK rows per call):
  // the x gather address is LINEAR in (nb, oh, ow), so one carry loop
  // replaces the two 64-bit div/mod chains per row per tile.
  const long strideC = (long)sh.stride * sh.C;
  const long strideWC = (long)sh.stride * sh.W * sh.C;
  const long HWC = (long)sh.H * sh.W * sh.C;
  int bow[2], boh[2];
  long baddr[2];  // x + baddr[j] = this row's (ih,iw) patch element
  {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      long k = k_lo + bmr + j * 16;
      bow[j] = (int)(k % sh.OW);
      boh[j] = (int)((k / sh.OW) % sh.OH);
      long nb = k / ((long)sh.OW * sh.OH);
      baddr[j] = nb * HWC +
                 ((long)(boh[j] * sh.stride - sh.pad + br_) * sh.W +
                  (bow[j] * sh.stride - sh.pad + bs_)) * sh.C + bc0_;
    }
  }
  auto b_advance = [&]() {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      bow[j] += BK;
      baddr[j] += BK * strideC;
      while (bow[j] >= sh.OW) {
        bow[j] -= sh.OW;
        baddr[j] += strideWC - (long)sh.OW * strideC;
        if (++boh[j] == sh.OH) {
          boh[j] = 0;
          baddr[j] += HWC - (long)sh.OH * strideWC;
        }
      }
    }
  };

  float raA[2][4];
  float4 rbB[2];
  auto stage_load = [&](long k0) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      long k = k0 + amr + j * 16;  // the m index
      float4 q = {0.f, 0.f, 0.f, 0.f};
      if (k < k_hi) {
        const float* src = dy + a_base + j * a_j16;
        if (a_v4) {
          q = *(const float4*)src;
        } else {
#pragma unroll
          for (int e = 0; e < 4; ++e)
            if (m_blk + ako + e < sh.Kout) (&q.x)[e] = src[e];
        }
      }
      raA[j][0] = q.x; raA[j][1] = q.y; raA[j][2] = q.z; raA[j][3] = q.w;
    }
    a_base += (long)BK * sh.Kout;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      long k = k0 + bmr + j * 16;
      float4 q = {0.f, 0.f, 0.f, 0.f};
      if (k < k_hi) {
        if (V4) {
          bool inb = P0;
          if (!P0) {
            int ih = boh[j] * sh.stride - sh.pad + br_;
            int iw = bow[j] * sh.stride - sh.pad + bs_;
            inb = (unsigned)ih < (unsigned)sh.H &&
                  (unsigned)iw < (unsigned)sh.W;
          }
          if (b_v4q && inb)
            q = *(const float4*)(x + baddr[j]);
          else if (b_any && inb) {
#pragma unroll
            for (int e = 0; e < 4; ++e)
              if (n_blk + bn4 + e < Ncrs) (&q.x)[e] = x[baddr[j] + e];
          }
        } else {
          int ow = bow[j], oh = boh[j];
          long xb = baddr[j];  // recover nb*HWC from the linear address
          xb -= ((long)(oh * sh.stride - sh.pad + br_) * sh.W +
                 (ow * sh.stride - sh.pad + bs_)) * sh.C + bc0_;
#pragma unroll
          for (int e = 0; e < 4; ++e) {
            int crs = n_blk + bn4 + e;
            if (crs < Ncrs) {
              int rs = crs / sh.C;
              int c = crs - rs * sh.C;
              int r = rs / sh.S, s = rs % sh.S;
              int ih = oh * sh.stride - sh.pad + r;
              int iw = ow * sh.stride - sh.pad + s;
              if (P0 || (ih >= 0 && ih < sh.H && iw >= 0 && iw < sh.W))
                (&q.x)[e] = x[xb + ((long)ih * sh.W + iw) * sh.C + c];
            }
          }
        }
      }
      rbB[j] = q;
    }
    b_advance();
  };
  auto stage_write = [&](int buf) {
    // A transposed: element (m, ko+i) -> A_lds[ko+i][m]
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int m = amr + j * 16;
#pragma unroll
      for (int i = 0; i < 4; ++i)
        A_lds[buf][(ako + i) * LDAW + m] = raA[j][i];
    }
#pragma unroll
    for (int j = 0; j < 2; ++j)
      *(float4*)&B_lds[buf][(bmr + j * 16) * LDB_S + bn4] = rbB[j];
  };

  stage_load(k_lo);
  stage_write(0);
  if (k_lo + BK < k_hi) stage_load(k_lo + BK);
  __syncthreads();
  int buf = 0;
  for (long k0 = k_lo; k0 < k_hi; k0 += BK) {
    if (k0 + BK < k_hi) {
      stage_write(buf ^ 1);
      if (k0 + 2 * BK < k_hi) stage_load(k0 + 2 * BK);
    }
    mfma_tile<MI, NI, LDAW>(A_lds[buf], B_lds[buf], wr, wc, l15, l4, acc);
    __syncthreads();
    buf ^= 1;
  }

  // A/B roles here: A rows = ko, B cols = (r,s,c)
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      int crs = n_blk + wc * 32 + ni * 16 + l15;
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

// ------------------------- small-C direct kernels (first layers) -------
// For C*R*S <= 32 the MFMA tile wastes most of its K (conv1: Kdim 9 padded
// to 32) and most of its N; a direct kernel at the VALU/memory roofline is
// several times faster.  Thread per output pixel m: all Kout accumulators
// in registers, weights staged once in LDS, NHWC float4 stores.

template <int KO_T>
__global__ __launch_bounds__(256)
void conv_small_fwd_k(const float* __restrict__ x,
                      const float* __restrict__ wt,  // [(r,s,c)][KO]
                      const float* __restrict__ bias, float* __restrict__ y,
                      ConvShape sh, int Kdim, int relu) {
  __shared__ float w_lds[32 * KO_T];
  for (int i = threadIdx.x; i < Kdim * sh.Kout; i += blockDim.x)
    w_lds[i] = wt[i];
  __syncthreads();
  long M = (long)sh.Nb * sh.OH * sh.OW;
  long stride = (long)gridDim.x * blockDim.x;
  for (long m = (long)blockIdx.x * blockDim.x + threadIdx.x; m < M;
       m += stride) {
    int ow = m % sh.OW;
    int oh = (m / sh.OW) % sh.OH;
    long nb = m / ((long)sh.OW * sh.OH);
    int oh0 = oh * sh.stride - sh.pad, ow0 = ow * sh.stride - sh.pad;
    const float* xp = x + nb * (long)sh.H * sh.W * sh.C;
    float acc[KO_T];
#pragma unroll
    for (int k = 0; k < KO_T; ++k) acc[k] = bias ? bias[k] : 0.f;
    for (int rs = 0; rs < Kdim / sh.C; ++rs) {
      int r = rs / sh.S, s = rs % sh.S;
      int ih = oh0 + r, iw = ow0 + s;
      if (ih < 0 || ih >= sh.H || iw < 0 || iw >= sh.W) continue;
      const float* xrow = xp + ((long)ih * sh.W + iw) * sh.C;
      for (int c = 0; c < sh.C; ++c) {
        float v = xrow[c];
        const float* wrow = &w_lds[(rs * sh.C + c) * KO_T];
#pragma unroll
        for (int k = 0; k < KO_T; ++k) acc[k] = fmaf(v, wrow[k], acc[k]);
      }
    }
    float* yo = y + m * sh.Kout;
    // float4 stores: the scalar form issued KO_T stores per pixel and the
    // kernel sat 4.5x over its write floor
#pragma unroll
    for (int k = 0; k < KO_T; k += 4) {
      float4 q = {acc[k], acc[k + 1], acc[k + 2], acc[k + 3]};
      if (relu) {
        q.x = fmaxf(q.x, 0.f); q.y = fmaxf(q.y, 0.f);
        q.z = fmaxf(q.z, 0.f); q.w = fmaxf(q.w, 0.f);
      }
      *(float4*)(yo + k) = q;
    }
  }
}

// dw partials for small Ncrs: block handles an m-chunk; per-thread (ko,crs)
// pairs accumulate from LDS-staged dy/x tiles; fixed-order combine follows.
// LDS is TRANSPOSED ([ko][m], [crs][m]) so the pairs loop reads float4
// over m — 4x fewer LDS instructions than the [m][ko] form; dy staging is
// a linear coalesced float4 copy (dy[m0*KO:] IS [m][ko] flat); x staging
// exploits that one m's (r, s, c) taps are R contiguous rows of S*C
// floats, with the m->(nb,oh,ow) decomposition carried incrementally.
template <int MC>
__global__ __launch_bounds__(256)
void conv_small_bwdw_k(const float* __restrict__ dy,
                       const float* __restrict__ x,
                       float* __restrict__ partials,  // [chunk][KO*Ncrs]
                       ConvShape sh, int Ncrs, long k_per_chunk) {
  constexpr int LDR = MC + 4;          // 16-B aligned row stride
  __shared__ float dy_lds[64][LDR];    // [ko][m] (KO <= 64)
  __shared__ float xp_lds[32][LDR];    // [crs][m] (Ncrs <= 32)
  const long Kdim = (long)sh.Nb * sh.OH * sh.OW;
  const long k_lo = (long)blockIdx.x * k_per_chunk;  // multiple of MC
  const long k_hi = min(Kdim, k_lo + k_per_chunk);
  const int KO = sh.Kout;
  const int pairs = KO * Ncrs;
  const int t = threadIdx.x;
  const int SC = sh.S * sh.C;
  const bool ko_p2 = (KO & (KO - 1)) == 0;
  const int kosh = 31 - __clz(KO);
  float acc[8];  // up to 8 pairs per thread (KO*Ncrs <= 2048)
#pragma unroll
  for (int i = 0; i < 8; ++i) acc[i] = 0.f;

  // this thread's m-row state (m = m0 + t), advanced by MC per chunk
  const bool mown = t < MC;
  int ow_t = 0, oh_t = 0;
  long nb_t = 0;
  if (mown) {
    long m = k_lo + t;
    ow_t = (int)(m % sh.OW);
    oh_t = (int)((m / sh.OW) % sh.OH);
    nb_t = m / ((long)sh.OW * sh.OH);
  }

  for (long m0 = k_lo; m0 < k_hi; m0 += MC) {
    int mc = (int)min((long)MC, k_hi - m0);
    {  // dy: linear float4 read, transposed scalar LDS writes
      const float* src = dy + m0 * KO;
      int t4 = mc * KO / 4;  // KO % 4 == 0
      for (int i = t; i < t4; i += blockDim.x) {
        float4 v = *(const float4*)(src + (long)i * 4);
        int e = i * 4;
        int m = ko_p2 ? (e >> kosh) : (e / KO);
        int ko = e - m * KO;
        dy_lds[ko + 0][m] = v.x;
        dy_lds[ko + 1][m] = v.y;
        dy_lds[ko + 2][m] = v.z;
        dy_lds[ko + 3][m] = v.w;
      }
      if (mc < MC)  // zero the tail rows so the full-MC pairs loop is exact
        for (int i = t; i < (MC - mc) * KO; i += blockDim.x)
          dy_lds[i % KO][mc + i / KO] = 0.f;
    }
    if (mown) {  // x patches for row m = m0 + t
      if (m0 + t < k_hi) {
        int ih0 = oh_t * sh.stride - sh.pad;
        int iw0 = ow_t * sh.stride - sh.pad;
        const float* xb = x + nb_t * (long)sh.H * sh.W * sh.C;
        int crs = 0;
        for (int r = 0; r < sh.R; ++r) {
          int ih = ih0 + r;
          const float* row = xb + ((long)ih * sh.W + iw0) * sh.C;
          bool rok = (unsigned)ih < (unsigned)sh.H;
          if (rok && sh.pad == 0) {
            for (int j = 0; j < SC; ++j) xp_lds[crs + j][t] = row[j];
          } else {
            for (int s2 = 0; s2 < sh.S; ++s2) {
              int iw = iw0 + s2;
              bool ok = rok && (unsigned)iw < (unsigned)sh.W;
              for (int c = 0; c < sh.C; ++c)
                xp_lds[crs + s2 * sh.C + c][t] =
                    ok ? row[s2 * sh.C + c] : 0.f;
            }
          }
          crs += SC;
        }
      } else {
        for (int j = 0; j < Ncrs; ++j) xp_lds[j][t] = 0.f;
      }
      ow_t += MC;
      while (ow_t >= sh.OW) {
        ow_t -= sh.OW;
        if (++oh_t == sh.OH) { oh_t = 0; ++nb_t; }
      }
    }
    __syncthreads();
    for (int pp = t, pi = 0; pp < pairs; pp += blockDim.x, ++pi) {
      int ko = pp / Ncrs, crs = pp - (pp / Ncrs) * Ncrs;
      const float4* dp = (const float4*)dy_lds[ko];
      const float4* xp = (const float4*)xp_lds[crs];
      float a = acc[pi];
#pragma unroll 4
      for (int m4 = 0; m4 < MC / 4; ++m4) {
        float4 d = dp[m4];
        float4 xv = xp[m4];
        a = fmaf(d.x, xv.x, a);
        a = fmaf(d.y, xv.y, a);
        a = fmaf(d.z, xv.z, a);
        a = fmaf(d.w, xv.w, a);
      }
      acc[pi] = a;
    }
    __syncthreads();
  }
  for (int p = t, pi = 0; p < pairs; p += blockDim.x, ++pi)
    partials[(long)blockIdx.x * pairs + p] = acc[pi];
}

// combine partials (fixed order, wave per pair) + permute (r,s,c)->(c,r,s)
__global__ void conv_small_bwdw_reduce_k(const float* __restrict__ partials,
                                         float* __restrict__ dw, int KO,
                                         int C, int RS, int chunks) {
  int pairs = KO * C * RS;
  int p = blockIdx.x * (blockDim.x / kWave) + threadIdx.x / kWave;
  int lane = threadIdx.x % kWave;
  if (p >= pairs) return;
  float acc = 0.f;
  for (int ch = lane; ch < chunks; ch += kWave)
    acc += partials[(long)ch * pairs + p];
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, kWave);
  if (lane == 0) {
    int crs = p % (C * RS);   // (r,s,c)
    int ko = p / (C * RS);
    int c = crs % C;
    int rs = crs / C;
    dw[(ko * C + c) * RS + rs] = acc;
  }
}

// fused split-K combine + dw permute: sums the SK rsc-ordered slabs and
// writes straight into the torch (ko,(c,r,s)) layout — one kernel instead
// of splitk_reduce + dwperm (the intermediate rsc tensor disappears)
__global__ void splitk_reduce_dwperm_k(const float* __restrict__ ws,
                                       float* __restrict__ dw, int Kout,
                                       int C, int RS, int SK) {
  long n_out = (long)Kout * C * RS;
  long stride = (long)gridDim.x * blockDim.x;
  int CRS = C * RS;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n_out;
       i += stride) {
    float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
    int z = 0;
    for (; z + 3 < SK; z += 4) {
      a0 += ws[(long)z * n_out + i];
      a1 += ws[(long)(z + 1) * n_out + i];
      a2 += ws[(long)(z + 2) * n_out + i];
      a3 += ws[(long)(z + 3) * n_out + i];
    }
    for (; z < SK; ++z) a0 += ws[(long)z * n_out + i];
    float acc = (a0 + a1) + (a2 + a3);
    int col = i % CRS;  // (r,s,c) flat: rs*C + c
    long ko = i / CRS;
    int c = col % C;
    int rs = col / C;
    dw[(ko * C + c) * RS + rs] = acc;
  }
}

// dw column permute: [(ko)][(r,s,c)] -> torch layout [(ko)][(c,r,s)]
__global__ void dwperm_rsc_crs_k(const float* __restrict__ in,
                                 float* __restrict__ out, int Kout, int C,
                                 int RS) {
  long n = (long)Kout * C * RS;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int col = i % (C * RS);   // (r,s,c) flat
    long ko = i / (C * RS);
    int c = col % C;
    int rs = col / C;
    out[(ko * C + c) * RS + rs] = in[i];
  }
}

// ---- conv bias gradient (NHWC): db[ko] = column sum of dy [M][KO] ----
// Dynamic chunk count keeps >=64k threads busy regardless of Kout; stage 2
// is one wave per ko over the chunk partials (shuffle tree, deterministic).
// thread -> (slice = tid / K_blk, ko = tid % K_blk): all four waves of a
// block work even when Kout < 256 (the one-slice form left 7/8 of each
// block idle at Kout=32 — 21.5 us for a 9.4 MB column sum)
__global__ void conv_db_stage1_k(const float* __restrict__ dy,
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
  for (long m = lo; m < hi; ++m) acc += dy[m * Kout + ko];
  partials[(long)chunk * Kout + ko] = acc;
}

__global__ void conv_db_stage2_k(const float* __restrict__ partials,
                                 float* __restrict__ db, int Kout,
                                 int chunks) {
  // one BLOCK per ko: 256 threads stride the chunks, LDS tree combine
  int ko = blockIdx.x;
  __shared__ float sh[kBlock];
  float acc = 0.f;
  for (int c = threadIdx.x; c < chunks; c += blockDim.x)
    acc += partials[(long)c * Kout + ko];
  sh[threadIdx.x] = acc;
  __syncthreads();
  for (int off = kBlock / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) sh[threadIdx.x] += sh[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) db[ko] = sh[0];
}

// permute w (Kout,C,R,S) -> staged layouts
__global__ void wperm_rsc_ko_k(const float* __restrict__ w,
                               float* __restrict__ out, int Kout, int C,
                               int RS) {
  // out[((rs)*C + c)*Kout + ko] = w[(ko*C + c)*RS + rs]
  long n = (long)Kout * C * RS;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int rs = i % RS;
    int c = (i / RS) % C;
    long ko = i / ((long)RS * C);
    out[((long)rs * C + c) * Kout + ko] = w[i];
  }
}

__global__ void wperm_rsko_c_k(const float* __restrict__ w,
                               float* __restrict__ out, int Kout, int C,
                               int RS) {
  // out[((rs)*Kout + ko)*C + c] = w[(ko*C + c)*RS + rs]
  long n = (long)Kout * C * RS;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int rs = i % RS;
    int c = (i / RS) % C;
    long ko = i / ((long)RS * C);
    out[((long)rs * Kout + ko) * C + c] = w[i];
  }
}

extern "C" void launch_splitk_reduce(const float* ws, float* C,
                                     const float* bias, int M, int N,
                                     int ldc, int SK, int relu, void* s);

extern "C" {
void launch_conv_fwd(const float* x, const float* wt, const float* bias,
                     float* y, int Nb, int C, int H, int W, int Kout, int R,
                     int S, int OH, int OW, int stride, int pad, int relu,
                     void* s) {
  ConvShape sh{Nb, C, H, W, Kout, R, S, OH, OW, stride, pad};
  int Kdim = C * R * S;
  long M = (long)Nb * OH * OW;
  dim3 grid((M + 127) / 128, (Kout + BN - 1) / BN, 1);
  hipStream_t st = (hipStream_t)s;
  if (Kdim <= 32 && (Kout == 32 || Kout == 64)) {
    if (Kout == 32)
      conv_small_fwd_k<32><<<grid_for(M), 256, 0, st>>>(x, wt, bias, y, sh,
                                                        Kdim, relu);
    else
      conv_small_fwd_k<64><<<grid_for(M), 256, 0, st>>>(x, wt, bias, y, sh,
                                                        Kdim, relu);
    return;
  }
  bool v4 = (C % 32) == 0;
  if (pad == 0 && stride == 1 && v4 && OH * OW >= 128) {
    // x-resident fast path: LDS = x region + 2 B buffers
    int ptiles = (OH * OW + 127) / 128;
    int oh_span = (127 / OW) + 2;               // output rows per tile (+1)
    int lds_rows = oh_span + R - 1;
    long xf = ((long)lds_rows * W * (C + 1) + 3) & ~3L;
    long lds_bytes = (xf + 2L * BK * LDB_S) * 4;
    if (lds_bytes <= 100 * 1024) {
      dim3 g2((long)Nb * ptiles, (Kout + BN - 1) / BN, 1);
      conv_fwd_res_k<<<g2, 256, lds_bytes, st>>>(x, wt, bias, y, sh, Kdim,
                                                 relu, ptiles, lds_rows);
      return;
    }
  }
  if (pad == 0 && v4)
    conv_fwd_k<true, true><<<grid, 256, 0, st>>>(x, wt, bias, y, sh, Kdim,
                                                 relu);
  else if (pad == 0)
    conv_fwd_k<true, false><<<grid, 256, 0, st>>>(x, wt, bias, y, sh, Kdim,
                                                  relu);
  else if (v4)
    conv_fwd_k<false, true><<<grid, 256, 0, st>>>(x, wt, bias, y, sh, Kdim,
                                                  relu);
  else
    conv_fwd_k<false, false><<<grid, 256, 0, st>>>(x, wt, bias, y, sh, Kdim,
                                                   relu);
}

void launch_conv_bwd_data_relu(const float* dy, const float* wp,
                               float* dx, const float* relu_y, int Nb,
                               int C, int H, int W, int Kout, int R, int S,
                               int OH, int OW, int stride, int pad,
                               void* s) {
  ConvShape sh{Nb, C, H, W, Kout, R, S, OH, OW, stride, pad};
  int Kdim = Kout * R * S;
  long M = (long)Nb * H * W;
  dim3 grid((M + 127) / 128, (C + BN - 1) / BN, 1);
  hipStream_t st = (hipStream_t)s;
  bool v4 = (Kout % 32) == 0;
  if (C <= 32) {
    dim3 g32((M + 127) / 128, (C + 31) / 32, 1);
    if (stride == 1 && v4)
      conv_bwd_data32_k<1, true><<<g32, 256, 0, st>>>(dy, wp, dx, sh, Kdim, relu_y);
    else if (stride == 2 && v4)
      conv_bwd_data32_k<2, true><<<g32, 256, 0, st>>>(dy, wp, dx, sh, Kdim, relu_y);
    else if (stride == 1)
      conv_bwd_data32_k<1, false><<<g32, 256, 0, st>>>(dy, wp, dx, sh,
                                                       Kdim, relu_y);
    else
      conv_bwd_data32_k<0, false><<<g32, 256, 0, st>>>(dy, wp, dx, sh,
                                                       Kdim, relu_y);
    return;
  }
  if (stride == 1 && v4)
    conv_bwd_data_k<1, true><<<grid, 256, 0, st>>>(dy, wp, dx, sh, Kdim, relu_y);
  else if (stride == 2 && v4)
    conv_bwd_data_k<2, true><<<grid, 256, 0, st>>>(dy, wp, dx, sh, Kdim, relu_y);
  else if (stride == 1)
    conv_bwd_data_k<1, false><<<grid, 256, 0, st>>>(dy, wp, dx, sh, Kdim, relu_y);
  else if (stride == 2)
    conv_bwd_data_k<2, false><<<grid, 256, 0, st>>>(dy, wp, dx, sh, Kdim, relu_y);
  else
    conv_bwd_data_k<0, false><<<grid, 256, 0, st>>>(dy, wp, dx, sh, Kdim, relu_y);
}

void launch_conv_bwd_data(const float* dy, const float* wp, float* dx,
                          int Nb, int C, int H, int W, int Kout, int R,
                          int S, int OH, int OW, int stride, int pad,
                          void* s) {
  launch_conv_bwd_data_relu(dy, wp, dx, nullptr, Nb, C, H, W, Kout, R, S,
                            OH, OW, stride, pad, s);
}

int conv_bwd_weight_splitk(int Kout, int Ncrs, long Kdim) {
  long tiles = ((Kout + 63) / 64) * (long)((Ncrs + BN - 1) / BN);
  if (tiles >= 2304 || Kdim <= 2 * BK) return 1;
  // target ~4608 blocks, cap SK at 192: the MI355X SK sweep
  // (profiles/r01_bwdw_micro.md f32 section) put the optimum at
  // 4.5k blocks for the CIFAR shapes and SK~192 for the FMNIST conv2
  long want = (4608 + tiles - 1) / tiles;
  long max_chunks = (Kdim + BK - 1) / BK;
  long sk = want < max_chunks ? want : max_chunks;
  return (int)(sk < 1 ? 1 : (sk > 192 ? 192 : sk));
}

// ws: SK*Kout*Ncrs + Kout*Ncrs floats (split-K slabs + rsc-ordered temp)
void launch_conv_bwd_weight(const float* dy, const float* x, float* dw,
                            float* ws, int SK, int Nb, int C, int H, int W,
                            int Kout, int R, int S, int OH, int OW,
                            int stride, int pad, void* s) {
  ConvShape sh{Nb, C, H, W, Kout, R, S, OH, OW, stride, pad};
  int Ncrs = C * R * S;
  long Kdim = (long)Nb * OH * OW;
  long k_per_chunk =
      SK == 1 ? Kdim : (((Kdim + SK - 1) / SK + BK - 1) / BK) * BK;
  dim3 grid((Kout + 63) / 64, (Ncrs + BN - 1) / BN, SK);
  hipStream_t st = (hipStream_t)s;
  if (Ncrs <= 32 && Kout <= 64) {
    // one 128-row tile per block where possible: 512 fixed chunks left the
    // chip at 2 blocks/CU (48 KB LDS, 4 waves) — finer chunking trades a
    // longer fixed-order reduce for 3x the resident parallelism
    int chunks = (int)min((Kdim + 127) / 128, (long)2048);
    if (chunks < 64) chunks = 64;
    long kpc = ((Kdim + chunks - 1) / chunks + 127) / 128 * 128;
    conv_small_bwdw_k<128><<<chunks, 256, 0, st>>>(dy, x, ws, sh, Ncrs,
                                                   kpc);
    int pairs = Kout * Ncrs;
    int wpb = kBlock / kWave;
    conv_small_bwdw_reduce_k<<<(pairs + wpb - 1) / wpb, kBlock, 0, st>>>(
        ws, dw, Kout, C, R * S, chunks);
    return;
  }
  float* slabs = ws;                         // SK * Kout * Ncrs
  float* rsc = ws + (long)SK * Kout * Ncrs;  // Kout * Ncrs, rsc order
  float* target = SK == 1 ? rsc : slabs;
  bool v4 = (C % 4) == 0;
  if (pad == 0 && v4)
    conv_bwd_weight_k<true, true><<<grid, 256, 0, st>>>(
        dy, x, target, sh, Ncrs, k_per_chunk, SK == 1);
  else if (pad == 0)
    conv_bwd_weight_k<true, false><<<grid, 256, 0, st>>>(
        dy, x, target, sh, Ncrs, k_per_chunk, SK == 1);
  else if (v4)
    conv_bwd_weight_k<false, true><<<grid, 256, 0, st>>>(
        dy, x, target, sh, Ncrs, k_per_chunk, SK == 1);
  else
    conv_bwd_weight_k<false, false><<<grid, 256, 0, st>>>(
        dy, x, target, sh, Ncrs, k_per_chunk, SK == 1);
  // fused combine+permute (for SK==1 it degenerates to the permute of
  // the directly-written rsc tensor); deep SK first folds 16-slab chunks
  // with a parallel stage-1 into the extra workspace slabs
  long n_out = (long)Kout * Ncrs;
  if (SK > 16) {
    extern void launch_splitk_partial(const float*, float*, long, int,
                                      int, void*);
    int chunks = (SK + 15) / 16;
    float* ws2 = slabs + (long)SK * n_out + n_out;  // after slabs + rsc
    launch_splitk_partial(slabs, ws2, n_out, SK, 16, s);
    splitk_reduce_dwperm_k<<<grid_for(n_out), kBlock, 0, st>>>(
        ws2, dw, Kout, C, R * S, chunks);
  } else {
    splitk_reduce_dwperm_k<<<grid_for(n_out), kBlock, 0, st>>>(
        SK == 1 ? rsc : slabs, dw, Kout, C, R * S, SK);
  }
}

int conv_db_chunks(long M, int Kout) {
  long want = 131072 / (Kout < 1 ? 1 : Kout);   // ~128k threads
  long cap = (M + 63) / 64;                     // >= 64 rows per chunk
  long c = want < cap ? want : cap;
  if (c < 64) c = 64;
  if (c > 4096) c = 4096;
  return (int)c;
}

static int conv_db_kblk(int Kout) {
  if (Kout >= kBlock) return kBlock;
  return (kBlock % Kout == 0) ? Kout : kBlock;
}

void launch_conv_db(const float* dy, float* db, float* partials, int Nb,
                    int Kout, int OHW, void* s) {
  hipStream_t st = (hipStream_t)s;
  long M = (long)Nb * OHW;
  int chunks = conv_db_chunks(M, Kout);
  int K_blk = conv_db_kblk(Kout);
  int sub_per = kBlock / K_blk;
  dim3 g1((chunks + sub_per - 1) / sub_per, (Kout + K_blk - 1) / K_blk);
  conv_db_stage1_k<<<g1, kBlock, 0, st>>>(dy, partials, M, Kout, chunks,
                                          K_blk);
  conv_db_stage2_k<<<Kout, kBlock, 0, st>>>(partials, db, Kout, chunks);
}

void launch_splitk_reduce_dwperm(const float* ws, float* dw, int Kout,
                                 int C, int RS, int SK, void* s) {
  long n_out = (long)Kout * C * RS;
  if (SK > 16) {  // caller provides the stage-1 slabs after slabs + rsc
    extern void launch_splitk_partial(const float*, float*, long, int, int,
                                      void*);
    int chunks = (SK + 15) / 16;
    float* ws2 = const_cast<float*>(ws) + (long)(SK + 1) * n_out;
    launch_splitk_partial(ws, ws2, n_out, SK, 16, s);
    splitk_reduce_dwperm_k<<<grid_for(n_out), kBlock, 0, (hipStream_t)s>>>(
        ws2, dw, Kout, C, RS, chunks);
    return;
  }
  splitk_reduce_dwperm_k<<<grid_for(n_out), kBlock, 0, (hipStream_t)s>>>(
      ws, dw, Kout, C, RS, SK);
}

void launch_dwperm_rsc_crs(const float* in, float* out, int Kout, int C,
                           int RS, void* s) {
  dwperm_rsc_crs_k<<<grid_for((long)Kout * C * RS), kBlock, 0,
                     (hipStream_t)s>>>(in, out, Kout, C, RS);
}

void launch_conv_db_stage2(const float* partials, float* db, int Kout,
                           int chunks, void* s) {
  conv_db_stage2_k<<<Kout, kBlock, 0, (hipStream_t)s>>>(partials, db, Kout,
                                                        chunks);
}

void launch_wperm_crs_ko(const float* w, float* out, int Kout, int C, int RS,
                         void* s) {
  wperm_rsc_ko_k<<<grid_for((long)Kout * C * RS), kBlock, 0,
                   (hipStream_t)s>>>(w, out, Kout, C, RS);
}
void launch_wperm_kors_c(const float* w, float* out, int Kout, int C, int RS,
                         void* s) {
  wperm_rsko_c_k<<<grid_for((long)Kout * C * RS), kBlock, 0,
                   (hipStream_t)s>>>(w, out, Kout, C, RS);
}
}
