// Tap-resident bf16 convolution kernels for the ResNet18 3x3 shapes —
// the family that replaced the implicit-GEMM forms wherever the input
// re-read per (r,s) tap (up to ~9x L2/L3 traffic) was the bound:
//   conv_bwdw_tap_bf16_k      3x3 s1 p1 bwd-weight, W in {8,16,32}
//   conv_tap_fwd_bf16_k       3x3 s1 p1 fwd AND bwd-data (tap-flipped)
//   conv_tap_fwd_w4_bf16_k    the 4x4-image fwd/bwd-data (packed pairs)
//   conv_tap_bwdd_s2_bf16_k   3x3 s2 p1 bwd-data (parity-masked rows)
//   conv_bwdw_tap_s2_bf16_k   3x3 s2 p1 bwd-weight (parity-split copies)
// plus the shared fixed-order partial combines.  Everything deterministic:
// no atomics, image-group partials summed in index order.
//
// First member (bwd-weight s1) in detail:
//
// The implicit-GEMM bwd-weight kernel re-reads x once per (r,s) tap and
// dy once per crs-tile: ~600 MB of L2/L3 traffic per ResNet layer, ~16x
// its MFMA floor (rocprofv3 r02 evidence).  This kernel streams BOTH
// tensors ONCE per (ko,c) tile: a block owns a 32x32 (ko,c) tile and a
// group of images; x rows pass through a sliding LDS ring holding THREE
// horizontally pre-shifted copies of each row (so every tap's 8-element
// fragment is an aligned ds_read_b128 — an unshifted layout would need
// misaligned LDS reads), and all NINE taps accumulate simultaneously into
// 9 MFMA accumulators per wave.  The zero halo baked into the ring edges
// IS the pad-1 semantics.  Per-image-group partial tiles are written in
// the final (ko,c,r,s) order and summed by a fixed-order combine
// (deterministic, no atomics).
//
//   grid = (Kout/32, C/32, S);  S image-groups bound the slab memory for
//   the deep 512-channel layers.
#include "common.h"

typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef short bf16x8 __attribute__((ext_vector_type(8)));

template <int LOGW, int KOT = 32>
__global__ __launch_bounds__(256)
void conv_bwdw_tap_bf16_k(const unsigned short* __restrict__ dy,
                          const unsigned short* __restrict__ x,
                          float* __restrict__ partials, int Nb, int C,
                          int H, int Kout, int n_per_block) {
  constexpr int W = 1 << LOGW;
  constexpr int RPT = 32 / W;         // rows per 32-pixel k-tile
  // PIPE: a two-tile ring + double-buffered dy lets tile t+1's staging
  // run beside tile t's MFMA with ONE barrier per tile.  Measured LOSS at
  // every W (W=32: 79 -> 101 us; the doubled ring costs a residency level
  // and the spare waves were already covering the barriers) — kept as a
  // compile-time branch with the small-ring two-barrier form as default.
  constexpr bool PIPE = false;
  constexpr int RING = PIPE ? 2 * RPT + 2 : RPT + 2;
  constexpr int RS = W + 16;          // in-row stride (multiple of 8)
  constexpr int CT = 32;
  constexpr int MI = KOT / 32;        // wave ko-fragments (KOT=64: 2
                                      // MFMA per tap per wave — halves
                                      // the x-staging cost per MFMA)
  // per-c row stride: multiple of 8 (16-B aligned fragment reads) with a
  // dword spread that de-conflicts the 16-lane b128 read groups
  constexpr int CSTRIDE = RING * RS + 8;

  __shared__ __align__(16) unsigned short x_lds[3][CT][CSTRIDE];
  __shared__ __align__(16) unsigned short dy_lds[PIPE ? 2 : 1][KOT][32 + 8];

  const int t = threadIdx.x;
  const int wave = t >> 6, lane = t & 63;
  const int wr = wave >> 1, wc = wave & 1;  // 2x2 over (ko, c)
  const int l15 = lane & 15, l4 = lane >> 4;
  const int ko0 = blockIdx.x * KOT;
  const int c0 = blockIdx.y * CT;
  const int n_lo = blockIdx.z * n_per_block;
  const int n_hi = min(Nb, n_lo + n_per_block);
  const int tiles = H / RPT;

  // zero once: stage writes never touch the halo positions, so the ring
  // edges stay zero for the whole block (pad-1 for free)
  for (int i = t; i < 3 * CT * CSTRIDE; i += 256)
    ((unsigned short*)x_lds)[i] = 0;
  __syncthreads();

  f32x4 acc[9][MI];
#pragma unroll
  for (int i = 0; i < 9; ++i)
#pragma unroll
    for (int mi = 0; mi < MI; ++mi) acc[i][mi] = {0.f, 0.f, 0.f, 0.f};

  // per-lane fragment decomposition of the pixel chunk l4*8 (W%8==0:
  // a chunk never crosses a row)
  const int mrow_l = (l4 * 8) >> LOGW;
  const int col_l = (l4 * 8) & (W - 1);

  // stage one x row (row in [-1, H]; outside -> zeros) into ring `slot`,
  // all three shifted copies: copy s stores x[row][col] at position
  // col + 1 - s + 8, so a read at aligned col0 + 8 yields x[col0 + s - 1]
  const int oct_per_row = W * CT / 8;
  auto stage_row = [&](int n, int row, int slot) {
    bool valid = (unsigned)row < (unsigned)H;
    for (int o = t; o < oct_per_row; o += 256) {
      int col = o >> 2;             // CT/8 = 4 octets per column
      int coct = (o & 3) * 8;
      bf16x8 q = {0, 0, 0, 0, 0, 0, 0, 0};
      if (valid)
        q = *(const bf16x8*)(x +
                             ((((long)n * H + row) << LOGW) + col) * C +
                             c0 + coct);
#pragma unroll
      for (int s = 0; s < 3; ++s) {
        int pos = slot * RS + col + 1 - s + 8;
#pragma unroll
        for (int e = 0; e < 8; ++e) x_lds[s][coct + e][pos] =
            ((const unsigned short*)&q)[e];
      }
    }
  };

  auto stage_dy = [&](int n, int trow0, int buf) {
    // 32 pixels x KOT/8 ko-octets, transposed to [ko][pixel]
    long pix0 = ((long)n * H + trow0) << LOGW;
    for (int o = t; o < 32 * KOT / 8; o += 256) {
      int pix = o / (KOT / 8);
      int koct = (o - pix * (KOT / 8)) * 8;
      bf16x8 q = *(const bf16x8*)(dy + (pix0 + pix) * Kout + ko0 + koct);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        dy_lds[buf][koct + e][pix] = ((const unsigned short*)&q)[e];
    }
  };

  auto mfma_tile = [&](int trow0, int buf) {
    bf16x8 a[MI];
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
      a[mi] = *(const bf16x8*)&dy_lds[buf][wr * (16 * MI) + mi * 16 + l15]
                                    [l4 * 8];
    const int c_lane = wc * 16 + l15;
#pragma unroll
    for (int r = 0; r < 3; ++r) {
      int row = trow0 + mrow_l + r - 1;
      int slot = (row + 1) % RING;
      const unsigned short* base = &x_lds[0][c_lane][slot * RS + col_l + 8];
#pragma unroll
      for (int s = 0; s < 3; ++s) {
        bf16x8 b = *(const bf16x8*)(base + (long)s * CT * CSTRIDE);
#pragma unroll
        for (int mi = 0; mi < MI; ++mi)
          acc[r * 3 + s][mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[mi], b, acc[r * 3 + s][mi], 0, 0, 0);
      }
    }
  };

  for (int n = n_lo; n < n_hi; ++n) {
    if (PIPE) {
      // warm: rows for tiles 0 AND 1, plus tile 0's dy
#pragma unroll
      for (int r = -1; r <= 2 * RPT; ++r)
        if (r <= H) stage_row(n, r, (r + 1) % RING);
      stage_dy(n, 0, 0);
      __syncthreads();
      for (int tile = 0; tile < tiles; ++tile) {
        int trow0 = tile * RPT;
        mfma_tile(trow0, tile & 1);
        // stage tile t+1's dy and rows WHILE t computes: the new rows'
        // ring slots belong to rows < t*RPT-1, which tile t never reads
        if (tile + 1 < tiles) {
          stage_dy(n, trow0 + RPT, (tile + 1) & 1);
#pragma unroll
          for (int r2 = 1; r2 <= RPT; ++r2) {
            int row = trow0 + RPT + r2;
            if (row <= H) stage_row(n, row, (row + 1) % RING);
          }
        }
        __syncthreads();
      }
    } else {
      // small-image form: ring holds one tile (+halo); two barriers
#pragma unroll
      for (int r = -1; r <= RPT; ++r) stage_row(n, r, (r + 1) % RING);
      for (int tile = 0; tile < tiles; ++tile) {
        int trow0 = tile * RPT;
        stage_dy(n, trow0, 0);
        __syncthreads();
        mfma_tile(trow0, 0);
        __syncthreads();  // drain reads before the ring advances
        if (tile + 1 < tiles) {
#pragma unroll
          for (int r2 = 1; r2 <= RPT; ++r2) {
            int row = trow0 + RPT + r2;
            stage_row(n, row, (row + 1) % RING);
          }
        }
      }
      __syncthreads();  // image boundary: ring refill starts clean
    }
  }

  // write this block's (ko,c) tile partials in final (ko,c,r,s) order
  long zbase = (long)blockIdx.z * Kout * C * 9;
#pragma unroll
  for (int rs = 0; rs < 9; ++rs)
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        int ko = ko0 + wr * (16 * MI) + mi * 16 + l4 * 4 + e;
        int c = c0 + wc * 16 + l15;
        partials[zbase + ((long)ko * C + c) * 9 + rs] = acc[rs][mi][e];
      }
}

// fixed-order combine: dw[(ko*C + c)*9 + rs] = sum over image groups.
// Two stages when the slab count is deep: a thread-per-output z-loop at
// S=256 is parallelism-starved (36 k threads, one load in flight each —
// 35.8 us measured for a 37 MB reduction); stage 1 gives every
// (output, 16-slab chunk) its own thread (coalesced, 16x the threads),
// stage 2 folds the chunks.
__global__ void bwdw_tap_combine_k(const float* __restrict__ partials,
                                   float* __restrict__ dw, long n_out,
                                   int S, int zstride) {
  long stride = (long)gridDim.x * blockDim.x;
  long total = n_out * ((S + zstride - 1) / zstride);
  for (long v = (long)blockIdx.x * blockDim.x + threadIdx.x; v < total;
       v += stride) {
    long i = v % n_out;
    int chunk = (int)(v / n_out);
    int z0 = chunk * zstride, z1 = min(S, z0 + zstride);
    float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
    int z = z0;
    for (; z + 3 < z1; z += 4) {
      a0 += partials[(long)z * n_out + i];
      a1 += partials[(long)(z + 1) * n_out + i];
      a2 += partials[(long)(z + 2) * n_out + i];
      a3 += partials[(long)(z + 3) * n_out + i];
    }
    for (; z < z1; ++z) a0 += partials[(long)z * n_out + i];
    dw[(long)chunk * n_out + i] = (a0 + a1) + (a2 + a3);
  }
}

extern "C" {

// 1 if the tap kernel covers this shape
int conv_bwdw_tap_ok(int C, int H, int W, int Kout, int R, int S,
                     int stride, int pad) {
  if (R != 3 || S != 3 || stride != 1 || pad != 1) return 0;
  if ((C % 32) || (Kout % 32)) return 0;
  if (W != 8 && W != 16 && W != 32) return 0;
  int rpt = 32 / W;
  return (H % rpt) == 0;
}

// number of image-group slabs (bounds ws to ~64 MB of fp32 partials)
int conv_bwdw_tap_slabs(int Nb, int C, int Kout) {
  long kc9 = (long)Kout * C * 9 * 4;
  long s = (64L << 20) / kc9;
  if (s < 1) s = 1;
  if (s > Nb) s = Nb;
  return (int)s;
}

void launch_conv_bwdw_tap_bf16(const unsigned short* dy,
                               const unsigned short* x, float* dw,
                               float* ws, int Nb, int C, int H, int W,
                               int Kout, void* st) {
  hipStream_t s = (hipStream_t)st;
  int S = conv_bwdw_tap_slabs(Nb, C, Kout);
  int G = (Nb + S - 1) / S;
  S = (Nb + G - 1) / G;
  if (W == 32) {
    dim3 grid(Kout / 32, C / 32, S);
    conv_bwdw_tap_bf16_k<5><<<grid, 256, 0, s>>>(dy, x, ws, Nb, C, H, Kout,
                                                 G);
  } else if (W == 16) {
    dim3 grid(Kout / 32, C / 32, S);
    conv_bwdw_tap_bf16_k<4><<<grid, 256, 0, s>>>(dy, x, ws, Nb, C, H, Kout,
                                                 G);
  } else {
    // KOT=64 A/B at W=8: staging per MFMA halves but the grid halves too
    // — 142 vs 134 us, starvation wins.  Stays at 32-wide ko tiles.
    dim3 grid(Kout / 32, C / 32, S);
    conv_bwdw_tap_bf16_k<3><<<grid, 256, 0, s>>>(dy, x, ws, Nb, C, H, Kout,
                                                 G);
  }
  long n_out = (long)Kout * C * 9;
  if (S > 16) {
    // stage 1 writes its chunk sums PAST the S slabs (ws is allocated
    // with 16 extra slabs), then stage 2 folds the chunks
    int chunks = (S + 15) / 16;
    float* ws2 = ws + (long)S * n_out;
    bwdw_tap_combine_k<<<grid_for(n_out * chunks), kBlock, 0, s>>>(
        ws, ws2, n_out, S, 16);
    bwdw_tap_combine_k<<<grid_for(n_out), kBlock, 0, s>>>(ws2, dw, n_out,
                                                          chunks, chunks);
  } else {
    bwdw_tap_combine_k<<<grid_for(n_out), kBlock, 0, s>>>(ws, dw, n_out, S,
                                                          S);
  }
}
}

// ---------------------------------------------------------------------
// Tap-resident bf16 conv FORWARD / stride-1 BACKWARD-DATA (3x3 s1 p1,
// W in {8,16,32}, Cin % 32 == 0, Cout % 32 == 0).
//
// The implicit-GEMM fwd/bwd-data kernels re-gather their input once per
// (r,s) tap (~9x traffic through L2/L3).  Here the contraction runs along
// the CHANNEL axis, so the input tile lives in LDS in its natural NHWC
// layout (c contiguous = the MFMA k fragment; the shifted taps merely
// select a different PIXEL row of the fragment — no alignment tricks
// needed), staged ONCE per (supertile, c-chunk).  A zero halo column/row
// ring provides pad-1 for free.  One kernel serves both directions:
// backward-data is the same correlation with the tap index flipped
// (w[2-r][2-s]) and the weight matrix already stored as [(r,s,ko)][C].
//
//   grid = (Cout/32, Nb); block streams its image's 128-pixel supertiles,
//   looping c-chunks of 32 with 9-tap MFMA accumulation in registers.
// ---------------------------------------------------------------------

template <int LOGW, int SPLIT, int G>
__global__ __launch_bounds__(256)
void conv_tap_fwd_bf16_k(const unsigned short* __restrict__ xin,
                         const unsigned short* __restrict__ wt,
                         const float* __restrict__ bias,
                         unsigned short* __restrict__ y,
                         const unsigned short* __restrict__ relu_y,
                         int Nb, int Cin, int H, int Cout, int relu,
                         int flip) {
  constexpr int W = 1 << LOGW;
  constexpr int ST = LOGW == 3 ? 64 : 128;  // pixels per supertile
  constexpr int NT = ST / 32;               // 32-pixel tiles per supertile
  constexpr int RPS = ST / W;               // rows per supertile
  constexpr int XROWS = RPS + 2;            // + halo
  constexpr int CP = 36;                    // c stride (bank spread)
  constexpr int COT = 32, CCH = 32;
  constexpr int TOT_W = 9 * CCH * COT / 8;  // w octets per chunk
  constexpr int PXR = XROWS * (W + 2) * 4;  // x octets per image slice
  constexpr int TOT_X = G * PXR;
  constexpr int NW = (TOT_W + 255) / 256;
  constexpr int NX = (TOT_X + 255) / 256;

  // G images share one block (small deep-layer images would otherwise
  // re-stage the w chunk once per image); SPLIT spreads a big image's
  // supertiles over gridDim.z.  The next chunk's w/x GLOBAL loads are
  // prefetched into registers DURING the MFMA phase (PMC r02: without
  // this the kernel sat 57-83% in SQ_WAIT_ANY on the staging loads).
  __shared__ __align__(16) unsigned short x_lds[G][XROWS * (W + 2)][CP];
  __shared__ __align__(16) unsigned short w_lds[9][COT][CP];

  const int t = threadIdx.x;
  const int wave = t >> 6, lane = t & 63;
  const int wr = wave >> 1, wc = wave & 1;
  const int l15 = lane & 15, l4 = lane >> 4;
  const int ko0 = blockIdx.x * COT;
  const int n_lo = blockIdx.y * G;
  const int P = H * W;
  const int n_super = (P + ST - 1) / ST;

  bf16x8 pw[NW], px[NX];

  auto prefetch = [&](int c0, int row0p) {
#pragma unroll
    for (int k = 0; k < NW; ++k) {
      int o = t + k * 256;
      if (o < TOT_W) {
        int rsc = o >> 2;
        int rs = rsc >> 5;
        int c = rsc & 31;
        int koct = (o & 3) * 8;
        pw[k] = *(const bf16x8*)(wt + ((long)rs * Cin + c0 + c) * Cout +
                                 ko0 + koct);
      }
    }
#pragma unroll
    for (int k = 0; k < NX; ++k) {
      int o = t + k * 256;
      bf16x8 q = {0, 0, 0, 0, 0, 0, 0, 0};
      if (o < TOT_X) {
        int g = o / PXR;
        int rem = o - g * PXR;
        int i = rem >> 2;
        int oc = rem & 3;
        int xr = i / (W + 2);
        int col = i - xr * (W + 2) - 1;
        int row = row0p - 1 + xr;
        int n = n_lo + g;
        bool ok = n < Nb && (unsigned)row < (unsigned)H &&
                  (unsigned)col < (unsigned)W;
        if (ok)
          q = *(const bf16x8*)(xin +
                               ((((long)n * H + row) << LOGW) + col) * Cin +
                               c0 + oc * 8);
      }
      px[k] = q;
    }
  };

  auto write_lds = [&]() {
#pragma unroll
    for (int k = 0; k < NW; ++k) {
      int o = t + k * 256;
      if (o < TOT_W) {
        int rsc = o >> 2;
        int rs = rsc >> 5;
        int c = rsc & 31;
        int koct = (o & 3) * 8;
        int rsd = flip ? 8 - rs : rs;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          w_lds[rsd][koct + e][c] = ((const unsigned short*)&pw[k])[e];
      }
    }
#pragma unroll
    for (int k = 0; k < NX; ++k) {
      int o = t + k * 256;
      if (o < TOT_X) {
        int g = o / PXR;
        int rem = o - g * PXR;
        int i = rem >> 2;
        int oc = rem & 3;
        // scalar stores: a CP=40 layout enabling b128 stores measured 7%
        // WORSE (the wider row hurt the read-side more than the stores)
#pragma unroll
        for (int e = 0; e < 8; ++e)
          x_lds[g][i][oc * 8 + e] = ((const unsigned short*)&px[k])[e];
      }
    }
  };

  const int st0 = blockIdx.z;
  if (st0 < n_super) prefetch(0, (st0 * ST) >> LOGW);
  for (int st = st0; st < n_super; st += SPLIT) {
    const int p0 = st * ST;
    const int row0 = p0 >> LOGW;
    f32x4 acc[G][NT];
#pragma unroll
    for (int g = 0; g < G; ++g)
#pragma unroll
      for (int i = 0; i < NT; ++i) acc[g][i] = {0.f, 0.f, 0.f, 0.f};

    for (int c0 = 0; c0 < Cin; c0 += CCH) {
      write_lds();
      __syncthreads();
      // next chunk's (or next supertile's) global loads fly under the
      // MFMA phase below
      if (c0 + CCH < Cin)
        prefetch(c0 + CCH, row0);
      else if (st + SPLIT < n_super)
        prefetch(0, ((st + SPLIT) * ST) >> LOGW);

#pragma unroll
      for (int g = 0; g < G; ++g)
#pragma unroll
        for (int tt = 0; tt < NT; ++tt) {
          int p = tt * 32 + wr * 16 + l15;
          int prow = (p >> LOGW) + 1;
          int pcol = (p & (W - 1)) + 1;
          bool pval = p0 + p < P;
          int base = pval ? (prow * (W + 2) + pcol) : (1 * (W + 2) + 1);
#pragma unroll
          for (int r = 0; r < 3; ++r)
#pragma unroll
            for (int s2 = 0; s2 < 3; ++s2) {
              bf16x8 a = *(const bf16x8*)&x_lds[g][base +
                                                (r - 1) * (W + 2) +
                                                (s2 - 1)][l4 * 8];
              bf16x8 b =
                  *(const bf16x8*)&w_lds[r * 3 + s2][wc * 16 + l15][l4 * 8];
              acc[g][tt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  a, b, acc[g][tt], 0, 0, 0);
            }
        }
      __syncthreads();
    }

#pragma unroll
    for (int g = 0; g < G; ++g) {
      int n = n_lo + g;
      if (n >= Nb) continue;
#pragma unroll
      for (int tt = 0; tt < NT; ++tt)
#pragma unroll
        for (int e = 0; e < 4; ++e) {
          int p = p0 + tt * 32 + wr * 16 + l4 * 4 + e;
          int ko = ko0 + wc * 16 + l15;
          if (p >= P) continue;
          long oidx = ((long)n * P + p) * Cout + ko;
          float v = acc[g][tt][e];
          if (bias) v += bias[ko];
          if (relu) v = fmaxf(v, 0.f);
          if (relu_y && bf2f_(relu_y[oidx]) <= 0.f) v = 0.f;
          y[oidx] = f2bf_(v);
        }
    }
  }
}

extern "C" {

int conv_tap_fwd_ok(int Cin, int H, int W, int Cout, int R, int S,
                    int stride, int pad) {
  if (R != 3 || S != 3 || stride != 1 || pad != 1) return 0;
  if ((Cin % 32) || (Cout % 32)) return 0;
  return (W == 8 || W == 16 || W == 32) && H == W;
}

void launch_conv_tap_fwd_bf16(const unsigned short* x,
                              const unsigned short* wt, const float* bias,
                              unsigned short* y,
                              const unsigned short* relu_y, int Nb,
                              int Cin, int H, int W, int Cout, int relu,
                              int flip, void* st) {
  hipStream_t s = (hipStream_t)st;
  if (W == 32) {  // 8 supertiles/image: split 4-way for occupancy
    dim3 grid(Cout / 32, Nb, 4);
    conv_tap_fwd_bf16_k<5, 4, 1><<<grid, 256, 0, s>>>(
        x, wt, bias, y, relu_y, Nb, Cin, H, Cout, relu, flip);
  } else if (W == 16) {  // 2 supertiles/image: split 2-way
    dim3 grid(Cout / 32, Nb, 2);
    conv_tap_fwd_bf16_k<4, 2, 1><<<grid, 256, 0, s>>>(
        x, wt, bias, y, relu_y, Nb, Cin, H, Cout, relu, flip);
  } else {  // 64-pixel images: 2 images share a block's w staging
    dim3 grid(Cout / 32, (Nb + 1) / 2, 1);
    conv_tap_fwd_bf16_k<3, 1, 2><<<grid, 256, 0, s>>>(
        x, wt, bias, y, relu_y, Nb, Cin, H, Cout, relu, flip);
  }
}
}

// ---------------------------------------------------------------------
// Tap-resident bf16 STRIDE-2 backward-data (3x3 s2 p1, dx W in {8,16,32},
// KO % 32 == 0, C % 32 == 0): dx[ih][iw] = sum over taps (r,s) of
// dy[(ih+1-r)/2][(iw+1-s)/2] * w[r][s] where the division must be exact
// (parity): lanes whose pixel has the wrong parity for a tap read a
// dedicated zero LDS row, so every tap is still one MFMA.  dy is staged
// once per (supertile, ko-chunk) — the implicit-GEMM form re-gathers it
// per tap with stride/parity checks in the inner loop.
// ---------------------------------------------------------------------

template <int LOGW>
__global__ __launch_bounds__(256)
void conv_tap_bwdd_s2_bf16_k(const unsigned short* __restrict__ dy,
                             const unsigned short* __restrict__ wp,
                             unsigned short* __restrict__ dx,
                             const unsigned short* __restrict__ relu_y,
                             int Nb, int KO, int H, int C) {
  constexpr int W = 1 << LOGW;
  constexpr int OW = W / 2;
  constexpr int ST = LOGW == 3 ? 64 : 128;
  constexpr int NT = ST / 32;
  constexpr int RPS = ST / W;             // dx rows per supertile
  constexpr int DYR = RPS / 2 + 2;        // dy rows staged (+halo)
  constexpr int DYW = OW + 1;             // + one zero col for ow == OW
  constexpr int CP = 36;
  constexpr int COT = 32, KCH = 32;
  constexpr int ZROW = DYR * DYW;         // always-zero position

  __shared__ __align__(16) unsigned short dy_lds[DYR * DYW + 1][CP];
  __shared__ __align__(16) unsigned short w_lds[9][COT][CP];

  const int t = threadIdx.x;
  const int wave = t >> 6, lane = t & 63;
  const int wr = wave >> 1, wc = wave & 1;
  const int l15 = lane & 15, l4 = lane >> 4;
  const int c0 = blockIdx.x * COT;        // dx channels
  const int n = blockIdx.y;
  const int P = H * W;
  const int OH = H / 2;
  const int n_super = (P + ST - 1) / ST;

  for (int st = blockIdx.z; st < n_super; st += gridDim.z) {
    const int p0 = st * ST;
    const int r0 = p0 >> LOGW;            // first dx row
    const int oh_base = (r0 - 1) >> 1;    // first staged dy row (may be -1)
    f32x4 acc[NT];
#pragma unroll
    for (int i = 0; i < NT; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

    for (int k0 = 0; k0 < KO; k0 += KCH) {
      // w chunk: [(rs*KO + ko)][C] -> [rs][c][ko]
      for (int o = t; o < 9 * KCH * COT / 8; o += 256) {
        int rsk = o >> 2;                 // (rs, ko) pair; 4 c-octets
        int rs = rsk >> 5;
        int ko = rsk & 31;
        int coct = (o & 3) * 8;
        const unsigned short* src =
            wp + ((long)rs * KO + k0 + ko) * C + c0 + coct;
        // transposed write: w_lds[rs][c][ko]
#pragma unroll
        for (int e = 0; e < 8; ++e)
          w_lds[rs][coct + e][ko] = src[e];
      }
      // dy rows oh_base .. oh_base+DYR-1, col ow in [0, OW) (+1 zero col)
      for (int i = t; i < DYR * DYW; i += 256) {
        int lr = i / DYW;
        int ow = i - lr * DYW;
        int oh = oh_base + lr;
        bool ok = (unsigned)oh < (unsigned)OH && ow < OW;
        const unsigned short* src =
            dy + (((long)n * OH + (ok ? oh : 0)) * OW + (ok ? ow : 0)) *
                     KO + k0;
#pragma unroll
        for (int oc = 0; oc < 4; ++oc) {
          bf16x8 q = {0, 0, 0, 0, 0, 0, 0, 0};
          if (ok) q = *(const bf16x8*)(src + oc * 8);
#pragma unroll
          for (int e = 0; e < 8; ++e)
            dy_lds[i][oc * 8 + e] = ((const unsigned short*)&q)[e];
        }
      }
      if (t < CP)  // the zero row
#pragma unroll
        for (int e = 0; e < 1; ++e) dy_lds[ZROW][t] = 0;
      __syncthreads();

#pragma unroll
      for (int tt = 0; tt < NT; ++tt) {
        int p = tt * 32 + wr * 16 + l15;
        int ih = r0 + (p >> LOGW);
        int iw = p & (W - 1);
        bool pval = p0 + p < P;
#pragma unroll
        for (int r = 0; r < 3; ++r) {
          int nh = ih + 1 - r;
          int lr = (nh >> 1) - oh_base;
          bool okh = pval && !(nh & 1);
#pragma unroll
          for (int s2 = 0; s2 < 3; ++s2) {
            int nw = iw + 1 - s2;
            int ow = nw >> 1;
            bool ok = okh && !(nw & 1) && nw >= 0;
            int pos = ok ? lr * DYW + ow : ZROW;
            bf16x8 a = *(const bf16x8*)&dy_lds[pos][l4 * 8];
            bf16x8 b = *(const bf16x8*)&w_lds[r * 3 + s2][wc * 16 + l15]
                                             [l4 * 8];
            acc[tt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b,
                                                              acc[tt], 0,
                                                              0, 0);
          }
        }
      }
      __syncthreads();
    }

#pragma unroll
    for (int tt = 0; tt < NT; ++tt)
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        int p = p0 + tt * 32 + wr * 16 + l4 * 4 + e;
        int c = c0 + wc * 16 + l15;
        if (p >= P) continue;
        long oidx = ((long)n * P + p) * C + c;
        float v = acc[tt][e];
        if (relu_y && bf2f_(relu_y[oidx]) <= 0.f) v = 0.f;
        dx[oidx] = f2bf_(v);
      }
  }
}

extern "C" {

int conv_tap_bwdd_s2_ok(int C, int H, int W, int KO, int R, int S,
                        int stride, int pad) {
  if (R != 3 || S != 3 || stride != 2 || pad != 1) return 0;
  if ((C % 32) || (KO % 32)) return 0;
  return (W == 8 || W == 16 || W == 32) && H == W;
}

void launch_conv_tap_bwdd_s2_bf16(const unsigned short* dy,
                                  const unsigned short* wp,
                                  unsigned short* dx,
                                  const unsigned short* relu_y, int Nb,
                                  int KO, int H, int W, int C, void* st) {
  hipStream_t s = (hipStream_t)st;
  if (W == 32) {
    dim3 grid(C / 32, Nb, 4);
    conv_tap_bwdd_s2_bf16_k<5><<<grid, 256, 0, s>>>(dy, wp, dx, relu_y,
                                                    Nb, KO, H, C);
  } else if (W == 16) {
    dim3 grid(C / 32, Nb, 2);
    conv_tap_bwdd_s2_bf16_k<4><<<grid, 256, 0, s>>>(dy, wp, dx, relu_y,
                                                    Nb, KO, H, C);
  } else {
    dim3 grid(C / 32, Nb, 1);
    conv_tap_bwdd_s2_bf16_k<3><<<grid, 256, 0, s>>>(dy, wp, dx, relu_y,
                                                    Nb, KO, H, C);
  }
}
}

// ---------------------------------------------------------------------
// Tap-accumulator bf16 STRIDE-2 backward-weight (3x3 s2 p1, dy OW in
// {8,16}, C % 32 == 0, Kout % 32 == 0): dw[ko][c][r][s] = sum over dy
// pixels of dy[oh][ow][ko] * x[2oh+r-1][2ow+s-1][c].  The x columns a
// tap touches are parity-split: col 2ow (s=1) lives in an even-column
// copy, cols 2ow -/+ 1 (s=0/2) in two shifted odd-column copies — each
// OW wide, so every tap fragment is an aligned contiguous b128 read.
// Rows likewise: x row 2oh in an even-row record, 2oh+1 in an odd-row
// record (tap r=0 at oh reads the odd record of oh-1).  dy and x stream
// ONCE per (ko,c) tile like the stride-1 tap kernel.
// ---------------------------------------------------------------------

template <int LOGOW>
__global__ __launch_bounds__(256)
void conv_bwdw_tap_s2_bf16_k(const unsigned short* __restrict__ dy,
                             const unsigned short* __restrict__ x,
                             float* __restrict__ partials, int Nb, int C,
                             int OH, int Kout, int n_per_block) {
  constexpr int OW = 1 << LOGOW;
  constexpr int RPT = 32 / OW;        // dy rows per 32-pixel k-tile
  constexpr int RING = RPT + 2;       // oh-keyed ring (needs oh-1)
  constexpr int RS = OW + 16;         // in-row stride (multiple of 8)
  constexpr int KOT = 32, CT = 32;
  constexpr int CSTRIDE = RING * RS + 8;
  // copies: 0 = even cols shifted for s=0 (store x[2k+1] at k+1... see
  // stage_rows), 1 = even-col (s=1), 2 = odd-col (s=2); records: even
  // x-row (r=1) and odd x-row (r=0 at oh-1 / r=2 at oh)
  __shared__ __align__(16) unsigned short xe_lds[3][CT][CSTRIDE];  // even
  __shared__ __align__(16) unsigned short xo_lds[3][CT][CSTRIDE];  // odd
  __shared__ __align__(16) unsigned short dy_lds[KOT][32 + 8];

  const int t = threadIdx.x;
  const int wave = t >> 6, lane = t & 63;
  const int wr = wave >> 1, wc = wave & 1;
  const int l15 = lane & 15, l4 = lane >> 4;
  const int ko0 = blockIdx.x * KOT;
  const int c0 = blockIdx.y * CT;
  const int n_lo = blockIdx.z * n_per_block;
  const int n_hi = min(Nb, n_lo + n_per_block);
  const int tiles = OH / RPT;
  const int H = OH * 2, W = OW * 2;

  for (int i = t; i < 3 * CT * CSTRIDE; i += 256) {
    ((unsigned short*)xe_lds)[i] = 0;
    ((unsigned short*)xo_lds)[i] = 0;
  }
  __syncthreads();

  f32x4 acc[9];
#pragma unroll
  for (int i = 0; i < 9; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  const int mrow_l = (l4 * 8) >> LOGOW;
  const int col_l = (l4 * 8) & (OW - 1);

  // stage both x rows of dy-row `oh` (x rows 2oh and 2oh+1) into ring
  // slot (oh+1)%RING.  Copy layout: a read at aligned position col0+8 of
  // copy s yields x[2*col0 + s - 1].
  const int oct_per = OW * CT / 8;    // octets per (row, col-parity)
  auto stage_rows = [&](int n, int oh, int slot) {
    // tasks: 2 x-rows (2oh, 2oh+1) x 2 column parities x OW/8*4 octets.
    // Copy contract: a read at position p+8 of copy cpy (in the record
    // matching the row parity) yields x[2p + cpy - 1]:
    //   even cols x[2k]   -> copy 1 at p = k
    //   odd  cols x[2k+1] -> copy 0 at p = k+1  AND  copy 2 at p = k
    for (int o = t; o < 4 * oct_per; o += 256) {
      int rowpar = o / (2 * oct_per);
      int rem = o - rowpar * (2 * oct_per);
      int colpar = rem >= oct_per;
      int oo = colpar ? rem - oct_per : rem;
      int k = oo >> 2;
      int coct = (oo & 3) * 8;
      int xrow = 2 * oh + rowpar;
      int xcol = 2 * k + colpar;
      bool ok = (unsigned)xrow < (unsigned)H;
      bf16x8 q = {0, 0, 0, 0, 0, 0, 0, 0};
      if (ok)
        q = *(const bf16x8*)(x + (((long)n * H + xrow) * W + xcol) * C +
                             c0 + coct);
      unsigned short* base = rowpar ? &xo_lds[0][0][0] : &xe_lds[0][0][0];
#pragma unroll
      for (int w2 = 0; w2 < 2; ++w2) {
        int cpy, p;
        if (colpar == 0) {
          if (w2 == 1) continue;
          cpy = 1; p = k;
        } else {
          cpy = w2 == 0 ? 0 : 2;
          p = w2 == 0 ? k + 1 : k;
        }
        unsigned short* dst = base + (long)cpy * CT * CSTRIDE;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          dst[(long)(coct + e) * CSTRIDE + slot * RS + p + 8] =
              ((const unsigned short*)&q)[e];
      }
    }
  };

  auto stage_dy = [&](int n, int trow0) {
    long pix0 = ((long)n * OH + trow0) << LOGOW;
    for (int o = t; o < 128; o += 256) {
      int pix = o >> 2;
      int koct = (o & 3) * 8;
      bf16x8 q = *(const bf16x8*)(dy + (pix0 + pix) * Kout + ko0 + koct);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        dy_lds[koct + e][pix] = ((const unsigned short*)&q)[e];
    }
  };

  for (int n = n_lo; n < n_hi; ++n) {
    // warm: dy rows -1 (odd record only matters) .. RPT
#pragma unroll
    for (int r = -1; r <= RPT; ++r) stage_rows(n, r, (r + 1) % RING);
    for (int tile = 0; tile < tiles; ++tile) {
      int trow0 = tile * RPT;
      stage_dy(n, trow0);
      __syncthreads();
      bf16x8 a = *(const bf16x8*)&dy_lds[wr * 16 + l15][l4 * 8];
      const int c_lane = wc * 16 + l15;
#pragma unroll
      for (int r = 0; r < 3; ++r) {
        // r=1: even record of oh; r=0: odd record of oh-1; r=2: odd of oh
        int oh = trow0 + mrow_l + (r == 0 ? -1 : 0);
        int slot = (oh + 1) % RING;
        const unsigned short* rec =
            (r == 1 ? &xe_lds[0][c_lane][0] : &xo_lds[0][c_lane][0]) +
            slot * RS + col_l + 8;
#pragma unroll
        for (int s2 = 0; s2 < 3; ++s2) {
          bf16x8 b = *(const bf16x8*)(rec + (long)s2 * CT * CSTRIDE);
          acc[r * 3 + s2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a, b, acc[r * 3 + s2], 0, 0, 0);
        }
      }
      __syncthreads();
      if (tile + 1 < tiles) {
#pragma unroll
        for (int r2 = 1; r2 <= RPT; ++r2) {
          int oh = trow0 + RPT + r2;
          stage_rows(n, oh, (oh + 1) % RING);
        }
      }
    }
    __syncthreads();
  }

  long zbase = (long)blockIdx.z * Kout * C * 9;
#pragma unroll
  for (int rs = 0; rs < 9; ++rs)
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      int ko = ko0 + wr * 16 + l4 * 4 + e;
      int c = c0 + wc * 16 + l15;
      partials[zbase + ((long)ko * C + c) * 9 + rs] = acc[rs][e];
    }
}

extern "C" {

int conv_bwdw_tap_s2_ok(int C, int H, int W, int Kout, int R, int S,
                        int stride, int pad) {
  if (R != 3 || S != 3 || stride != 2 || pad != 1) return 0;
  if ((C % 32) || (Kout % 32)) return 0;
  int OW = W / 2;
  // OW=16 only: at OW=8 the parity-split staging outweighs the traffic
  // win (tap 159 us vs ~130 implicit-GEMM, profiles/r02_final_resnet.md)
  if (OW != 16) return 0;
  int rpt = 32 / OW;
  return H == W && (H / 2) % rpt == 0;
}

void launch_conv_bwdw_tap_s2_bf16(const unsigned short* dy,
                                  const unsigned short* x, float* dw,
                                  float* ws, int Nb, int C, int H, int W,
                                  int Kout, void* st) {
  hipStream_t s = (hipStream_t)st;
  int S = conv_bwdw_tap_slabs(Nb, C, Kout);
  int G = (Nb + S - 1) / S;
  S = (Nb + G - 1) / G;
  dim3 grid(Kout / 32, C / 32, S);
  int OH = H / 2;
  if (W / 2 == 16)
    conv_bwdw_tap_s2_bf16_k<4><<<grid, 256, 0, s>>>(dy, x, ws, Nb, C, OH,
                                                    Kout, G);
  else
    conv_bwdw_tap_s2_bf16_k<3><<<grid, 256, 0, s>>>(dy, x, ws, Nb, C, OH,
                                                    Kout, G);
  long n_out = (long)Kout * C * 9;
  if (S > 16) {
    int chunks = (S + 15) / 16;
    float* ws2 = ws + (long)S * n_out;
    bwdw_tap_combine_k<<<grid_for(n_out * chunks), kBlock, 0, s>>>(
        ws, ws2, n_out, S, 16);
    bwdw_tap_combine_k<<<grid_for(n_out), kBlock, 0, s>>>(ws2, dw, n_out,
                                                          chunks, chunks);
  } else {
    bwdw_tap_combine_k<<<grid_for(n_out), kBlock, 0, s>>>(ws, dw, n_out, S,
                                                          S);
  }
}
}

// ---------------------------------------------------------------------
// W=4 tap-resident bf16 fwd / s1 bwd-data: 16-pixel images are PACKED
// two-per-32-pixel MFMA tile (the result-row dimension indexes a virtual
// pixel space across GI images whose LDS slices are contiguous).  Same
// channel-axis contraction and flip convention as conv_tap_fwd_bf16_k.
// ---------------------------------------------------------------------

template <int GI>  // images per block, multiple of 2
__global__ __launch_bounds__(256)
void conv_tap_fwd_w4_bf16_k(const unsigned short* __restrict__ xin,
                            const unsigned short* __restrict__ wt,
                            const float* __restrict__ bias,
                            unsigned short* __restrict__ y,
                            const unsigned short* __restrict__ relu_y,
                            int Nb, int Cin, int Cout, int relu,
                            int flip) {
  constexpr int W = 4, H = 4, P = 16;
  constexpr int XW = W + 2, XROWS = H + 2;   // whole image + halo
  constexpr int NPOS = XROWS * XW;           // 36
  constexpr int NT = GI / 2;                 // 32-pixel tiles per block
  constexpr int CP = 36;
  constexpr int COT = 32, CCH = 32;

  __shared__ __align__(16) unsigned short x_lds[GI * NPOS][CP];
  __shared__ __align__(16) unsigned short w_lds[9][COT][CP];

  const int t = threadIdx.x;
  const int wave = t >> 6, lane = t & 63;
  const int wr = wave >> 1, wc = wave & 1;
  const int l15 = lane & 15, l4 = lane >> 4;
  const int ko0 = blockIdx.x * COT;
  const int n_lo = blockIdx.y * GI;

  f32x4 acc[NT];
#pragma unroll
  for (int i = 0; i < NT; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  for (int c0 = 0; c0 < Cin; c0 += CCH) {
    for (int o = t; o < 9 * CCH * COT / 8; o += 256) {
      int rsc = o >> 2;
      int rs = rsc >> 5;
      int c = rsc & 31;
      int koct = (o & 3) * 8;
      bf16x8 q = *(const bf16x8*)(
          wt + ((long)rs * Cin + c0 + c) * Cout + ko0 + koct);
      int rsd = flip ? 8 - rs : rs;
#pragma unroll
      for (int e = 0; e < 8; ++e)
        w_lds[rsd][koct + e][c] = ((const unsigned short*)&q)[e];
    }
    for (int i = t; i < GI * NPOS; i += 256) {
      int g = i / NPOS;
      int pos = i - g * NPOS;
      int xr = pos / XW;
      int col = pos - xr * XW - 1;
      int row = xr - 1;
      int n = n_lo + g;
      bool ok = n < Nb && (unsigned)row < (unsigned)H &&
                (unsigned)col < (unsigned)W;
      const unsigned short* src =
          xin + (((long)(ok ? n : 0) * H + (ok ? row : 0)) * W +
                 (ok ? col : 0)) * Cin + c0;
#pragma unroll
      for (int oc = 0; oc < 4; ++oc) {
        bf16x8 q = {0, 0, 0, 0, 0, 0, 0, 0};
        if (ok) q = *(const bf16x8*)(src + oc * 8);
#pragma unroll
        for (int e = 0; e < 8; ++e)
          x_lds[i][oc * 8 + e] = ((const unsigned short*)&q)[e];
      }
    }
    __syncthreads();

#pragma unroll
    for (int tt = 0; tt < NT; ++tt) {
      int p = wr * 16 + l15;              // virtual pixel in [0, 32)
      int g = tt * 2 + (p >> 4);
      int pix = p & 15;
      int base = g * NPOS + ((pix >> 2) + 1) * XW + (pix & 3) + 1;
#pragma unroll
      for (int r = 0; r < 3; ++r)
#pragma unroll
        for (int s2 = 0; s2 < 3; ++s2) {
          bf16x8 a = *(const bf16x8*)&x_lds[base + (r - 1) * XW +
                                            (s2 - 1)][l4 * 8];
          bf16x8 b =
              *(const bf16x8*)&w_lds[r * 3 + s2][wc * 16 + l15][l4 * 8];
          acc[tt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[tt],
                                                            0, 0, 0);
        }
    }
    __syncthreads();
  }

#pragma unroll
  for (int tt = 0; tt < NT; ++tt)
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      int p = wr * 16 + l4 * 4 + e;
      int g = tt * 2 + (p >> 4);
      int pix = p & 15;
      int n = n_lo + g;
      int ko = ko0 + wc * 16 + l15;
      if (n >= Nb) continue;
      long oidx = ((long)n * P + pix) * Cout + ko;
      float v = acc[tt][e];
      if (bias) v += bias[ko];
      if (relu) v = fmaxf(v, 0.f);
      if (relu_y && bf2f_(relu_y[oidx]) <= 0.f) v = 0.f;
      y[oidx] = f2bf_(v);
    }
}

extern "C" {

int conv_tap_fwd_w4_ok(int Cin, int H, int W, int Cout, int R, int S,
                       int stride, int pad) {
  if (R != 3 || S != 3 || stride != 1 || pad != 1) return 0;
  if ((Cin % 32) || (Cout % 32)) return 0;
  return W == 4 && H == 4;
}

void launch_conv_tap_fwd_w4_bf16(const unsigned short* x,
                                 const unsigned short* wt,
                                 const float* bias, unsigned short* y,
                                 const unsigned short* relu_y, int Nb,
                                 int Cin, int Cout, int relu, int flip,
                                 void* st) {
  // GI=8 A/B was a wash (77.2 vs 76.4 us) — stay at 4 images/block for
  // the wider grid
  dim3 grid(Cout / 32, (Nb + 3) / 4);
  conv_tap_fwd_w4_bf16_k<4><<<grid, 256, 0, (hipStream_t)st>>>(
      x, wt, bias, y, relu_y, Nb, Cin, Cout, relu, flip);
}
}
