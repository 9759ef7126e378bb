// BatchNorm2d fwd/bwd — NHWC (channels_last), the build's ResNet18
// extension (the reference has no BN, SURVEY.md §2b last row).
// x is a [M = N*H*W][C] row-major matrix; per-channel statistics are
// two-stage column sums (chunk partials -> combine), deterministic, no
// atomics.  Elementwise passes use a stride-multiple-of-C grid so each
// thread's channel is computed ONCE (launcher rounds the grid).
//
// Stage-1 parallelism (rocprofv3 evidence, profiles/): 512 row chunks, and
// when C < 256 a block carries 256/C chunk SLICES so all four waves work —
// consecutive lanes read consecutive channels (coalesced 128 B per wave
// for bf16 at C>=64).  The first-cut 64-chunk/one-wave version left 3/4 of
// each block idle and only 64 blocks on a 256-CU chip: 322 us for a 33 MB
// pass that bn_norm_k covers in 8 us.
#include "common.h"

constexpr int kBnChunksMax = 2048;  // partials allocation bound
// row chunks scale with M: the big stem/L1 activations want deep stage-1
// parallelism (at 512 chunks each thread walked 512 serial rows — the
// 23.8 us bwd-stats1 was ~2.2x its read floor); small deep-layer tensors
// waste stage-2 iterations on empty chunks.  Multiple of 16 so stage-1
// slice blocks tile exactly.
static __host__ __device__ inline int bn_chunks(long M) {
  long c = M / 16;
  if (c > kBnChunksMax) c = kBnChunksMax;
  if (c < 64) c = 64;
  return (int)(c & ~15L);
}

// ---- stage 1: partials[chunk][2C] = (sum x, sum x^2) over a row chunk ----
// grid: x = kBnChunks / slices_per_block, y = ceil(C / C_blk);
// thread -> (slice = tid / C_blk, c = y*C_blk + tid % C_blk).
template <typename T>
__global__ void bn_stats1_k(const T* __restrict__ x,
                            float* __restrict__ partials, long M, int C,
                            int C_blk) {
  int sub_per = blockDim.x / C_blk;
  int c = blockIdx.y * C_blk + threadIdx.x % C_blk;
  int sub = threadIdx.x / C_blk;
  int chunk = blockIdx.x * sub_per + sub;
  const int chunks = bn_chunks(M);
  if (c >= C || chunk >= chunks) return;
  long per = (M + chunks - 1) / chunks;
  long lo = (long)chunk * per, hi = min(M, lo + per);
  float s = 0.f, ss = 0.f;
#pragma unroll 4
  for (long m = lo; m < hi; ++m) {
    float v = ldv(&x[m * C + c]);
    s += v;
    ss += v * v;
  }
  partials[((long)chunk * 2) * C + c] = s;
  partials[((long)chunk * 2 + 1) * C + c] = ss;
}

// ---- stage 2: sums[c], sums[C+c] = fixed-order combine of the chunk
// partials.  One block covers 16 channels x 16 chunk slices (32-iteration
// strided loops), slices combined in LDS in fixed order (deterministic).
// A single-block variant at 512 chunks measured 87 us — this is ~3 us.
// The fwd finalize (mean/rstd + running stats) is folded in behind
// `count > 0` — one launch fewer per BN forward (20/step on ResNet18).
__global__ void bn_stats2_k(const float* __restrict__ partials,
                            float* __restrict__ sums, int C, int chunks,
                            float* __restrict__ copy0,
                            float* __restrict__ copy1, long count,
                            float momentum, float eps,
                            float* __restrict__ save_mean,
                            float* __restrict__ save_rstd,
                            float* __restrict__ running_mean,
                            float* __restrict__ running_var) {
  // one block per TWO channels, 128 chunk-slices each: at 512 chunks a
  // 16-slice form left each thread 32 serial loads on a 4-block grid —
  // 9.8 us of pure latency per call, 40 calls per ResNet step
  constexpr int CB = 2, SL = 128;  // CB*SL == blockDim.x == 256
  __shared__ float l_s[256], l_ss[256];
  int lc = threadIdx.x % CB;
  int c = blockIdx.x * CB + lc;
  int sub = threadIdx.x / CB;
  float s = 0.f, ss = 0.f;
  if (c < C) {
    for (int ch = sub; ch < chunks; ch += SL) {
      s += partials[((long)ch * 2) * C + c];
      ss += partials[((long)ch * 2 + 1) * C + c];
    }
  }
  l_s[threadIdx.x] = s;
  l_ss[threadIdx.x] = ss;
  __syncthreads();
  // fixed-order slice fold (deterministic): tree over the 128 slices
  for (int off = SL / 2; off > 0; off >>= 1) {
    if (sub < off) {
      l_s[threadIdx.x] += l_s[threadIdx.x + off * CB];
      l_ss[threadIdx.x] += l_ss[threadIdx.x + off * CB];
    }
    __syncthreads();
  }
  if (sub == 0 && c < C) {
    s = l_s[lc];
    ss = l_ss[lc];
    sums[c] = s;
    sums[C + c] = ss;
    // bwd: db = sum(dy), dw = sum(dy*xhat) — written here instead of two
    // D2D copies after the fact (12k extra launches per bench, profiles/)
    if (copy0) copy0[c] = s;
    if (copy1) copy1[c] = ss;
    if (count > 0) {  // fused fwd finalize
      float mean = s / count;
      float var = ss / count - mean * mean;
      var = fmaxf(var, 0.f);
      save_mean[c] = mean;
      save_rstd[c] = rsqrtf(var + eps);
      if (running_mean) {
        float unbiased = count > 1 ? var * count / (count - 1) : var;
        running_mean[c] =
            (1.f - momentum) * running_mean[c] + momentum * mean;
        running_var[c] =
            (1.f - momentum) * running_var[c] + momentum * unbiased;
      }
    }
  }
}

// (bn_finalize_k was folded into bn_stats2_k's count>0 branch)

__global__ void bn_eval_stats_k(const float* __restrict__ running_mean,
                                const float* __restrict__ running_var,
                                float* __restrict__ save_mean,
                                float* __restrict__ save_rstd, int C,
                                float eps) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  save_mean[c] = running_mean[c];
  save_rstd[c] = rsqrtf(running_var[c] + eps);
}

// ---- normalize: y = w[c]*(x-mean[c])*rstd[c] + b[c]; stride % C == 0 so
// each thread's channel is fixed across its grid-stride loop ----
template <typename T>
__global__ void bn_norm_k(const T* __restrict__ x,
                          const float* __restrict__ w,
                          const float* __restrict__ b,
                          const float* __restrict__ mean,
                          const float* __restrict__ rstd,
                          T* __restrict__ y, long n, int C, int relu) {
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  if (stride % C == 0) {  // launcher arranges this for power-of-two C
    int c = (int)(i0 % C);
    float wc = w[c] * rstd[c];
    float bc = b[c] - mean[c] * wc;
    for (long i = i0; i < n; i += stride) {
      float v = ldv(&x[i]) * wc + bc;
      stv(&y[i], relu ? fmaxf(v, 0.f) : v);
    }
  } else {
    for (long i = i0; i < n; i += stride) {
      int c = (int)(i % C);
      float v = w[c] * rstd[c] * (ldv(&x[i]) - mean[c]) + b[c];
      stv(&y[i], relu ? fmaxf(v, 0.f) : v);
    }
  }
}

// ---- bwd stage 1: per-channel sum(dy), sum(dy * xhat) ----
template <typename T>
__global__ void bn_bwd_stats1_k(const T* __restrict__ x,
                                const T* __restrict__ dy,
                                const float* __restrict__ mean,
                                const float* __restrict__ rstd,
                                float* __restrict__ partials, long M, int C,
                                int C_blk) {
  int sub_per = blockDim.x / C_blk;
  int c = blockIdx.y * C_blk + threadIdx.x % C_blk;
  int sub = threadIdx.x / C_blk;
  int chunk = blockIdx.x * sub_per + sub;
  const int chunks = bn_chunks(M);
  if (c >= C || chunk >= chunks) return;
  long per = (M + chunks - 1) / chunks;
  long lo = (long)chunk * per, hi = min(M, lo + per);
  float m_ = mean[c], rs = rstd[c];
  float sdy = 0.f, sdyx = 0.f;
#pragma unroll 4
  for (long m = lo; m < hi; ++m) {
    float g = ldv(&dy[m * C + c]);
    sdy += g;
    sdyx += g * (ldv(&x[m * C + c]) - m_) * rs;
  }
  partials[((long)chunk * 2) * C + c] = sdy;
  partials[((long)chunk * 2 + 1) * C + c] = sdyx;
}

// (bn_stats2_k combines these too: same [chunk][2C] layout)

// ---- bwd: dx = w*rstd*(dy - (db + xhat*dwdot)/Nc) ----
template <typename T>
__global__ void bn_bwd_dx_k(const T* __restrict__ x,
                            const T* __restrict__ dy,
                            const float* __restrict__ w,
                            const float* __restrict__ mean,
                            const float* __restrict__ rstd,
                            const float* __restrict__ stats,
                            T* __restrict__ dx, long n, int C,
                            float inv_count) {
  long i0 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  if (stride % C == 0) {
    int c = (int)(i0 % C);
    float wr = w[c] * rstd[c];
    float m_ = mean[c], rs = rstd[c];
    float db = stats[c], dwdot = stats[C + c];
    for (long i = i0; i < n; i += stride) {
      float xhat = (ldv(&x[i]) - m_) * rs;
      stv(&dx[i], wr * (ldv(&dy[i]) - inv_count * (db + xhat * dwdot)));
    }
  } else {
    for (long i = i0; i < n; i += stride) {
      int c = (int)(i % C);
      float xhat = (ldv(&x[i]) - mean[c]) * rstd[c];
      stv(&dx[i], w[c] * rstd[c] *
                      (ldv(&dy[i]) -
                       inv_count * (stats[c] + xhat * stats[C + c])));
    }
  }
}

static int bn_grid(long n, int C) {
  // blocks such that gridDim*256 % C == 0 (C is a power of two <= 512 in
  // practice; the rounding below also handles other C by rounding up to a
  // multiple of C)
  long want = (n + kBlock - 1) / kBlock;
  if (want > kMaxBlocks) want = kMaxBlocks;
  long mult = (C + kBlock - 1) / kBlock;  // blocks per C-span
  long lcm = mult > 1 ? mult : 1;
  if (256 % C != 0 && C % 256 == 0) lcm = C / 256;
  if (lcm > 1) want = ((want + lcm - 1) / lcm) * lcm;
  return (int)(want > 0 ? want : 1);
}

// channels per block for the stats kernels: all of C up to the block size,
// and a divisor of 256 so slices tile the block exactly (non-power-of-two
// C falls back to one slice per block).
static int bn_cblk(int C) {
  if (C >= 256) return 256;
  return (256 % C == 0) ? C : 256;
}

template <typename T>
static void bn_fwd_impl(const T* x, const float* w, const float* b,
                        float* running_mean, float* running_var,
                        float* save_mean, float* save_rstd, T* y,
                        float* scratch, int Nb, int C, int HW,
                        float momentum, float eps, int training, int relu,
                        hipStream_t st) {
  long M = (long)Nb * HW;
  float* partials = scratch;
  float* sums = scratch + (long)kBnChunksMax * 2 * C;
  if (training) {
    int C_blk = bn_cblk(C);
    int sub_per = 256 / C_blk;
    int chunks = bn_chunks(M);
    dim3 g1((chunks + sub_per - 1) / sub_per, (C + C_blk - 1) / C_blk);
    bn_stats1_k<T><<<g1, 256, 0, st>>>(x, partials, M, C, C_blk);
    bn_stats2_k<<<(C + 1) / 2, 256, 0, st>>>(
        partials, sums, C, chunks, nullptr, nullptr, M, momentum, eps,
        save_mean, save_rstd, running_mean, running_var);
  } else {
    bn_eval_stats_k<<<(C + 255) / 256, 256, 0, st>>>(
        running_mean, running_var, save_mean, save_rstd, C, eps);
  }
  long n = M * C;
  bn_norm_k<T><<<bn_grid(n, C), kBlock, 0, st>>>(x, w, b, save_mean,
                                                 save_rstd, y, n, C, relu);
}

template <typename T>
static void bn_bwd_impl(const T* x, const T* dy, const float* w,
                        const float* save_mean, const float* save_rstd,
                        float* scratch, T* dx, float* dw, float* db, int Nb,
                        int C, int HW, int training, hipStream_t st) {
  long M = (long)Nb * HW;
  float* partials = scratch;
  float* stats = scratch + (long)kBnChunksMax * 2 * C;
  int C_blk = bn_cblk(C);
  int sub_per = 256 / C_blk;
  int chunks = bn_chunks(M);
  dim3 g1((chunks + sub_per - 1) / sub_per, (C + C_blk - 1) / C_blk);
  bn_bwd_stats1_k<T><<<g1, 256, 0, st>>>(x, dy, save_mean, save_rstd,
                                         partials, M, C, C_blk);
  bn_stats2_k<<<(C + 1) / 2, 256, 0, st>>>(partials, stats, C, chunks, db,
                                           dw, 0, 0.f, 0.f, nullptr,
                                           nullptr, nullptr, nullptr);
  long n = M * C;
  bn_bwd_dx_k<T><<<bn_grid(n, C), kBlock, 0, st>>>(
      x, dy, w, save_mean, save_rstd, stats, dx, n, C,
      training ? 1.f / (float)M : 0.f);
}

extern "C" {
int bn_scratch_floats(int C) { return (kBnChunksMax * 2 + 2) * C; }

void launch_bn_fwd(const float* x, const float* w, const float* b,
                   float* running_mean, float* running_var, float* save_mean,
                   float* save_rstd, float* y, float* scratch, int Nb,
                   int C, int HW, float momentum, float eps, int training,
                   int relu, void* s) {
  bn_fwd_impl<float>(x, w, b, running_mean, running_var, save_mean,
                     save_rstd, y, scratch, Nb, C, HW, momentum, eps,
                     training, relu, (hipStream_t)s);
}
void launch_bn_fwd_bf16(const unsigned short* x, const float* w,
                        const float* b, float* running_mean,
                        float* running_var, float* save_mean,
                        float* save_rstd, unsigned short* y, float* scratch,
                        int Nb, int C, int HW, float momentum, float eps,
                        int training, int relu, void* s) {
  bn_fwd_impl<unsigned short>(x, w, b, running_mean, running_var, save_mean,
                              save_rstd, y, scratch, Nb, C, HW, momentum,
                              eps, training, relu, (hipStream_t)s);
}
void launch_bn_bwd(const float* x, const float* dy, const float* w,
                   const float* save_mean, const float* save_rstd,
                   float* scratch, float* dx, float* dw, float* db, int Nb,
                   int C, int HW, int training, void* s) {
  bn_bwd_impl<float>(x, dy, w, save_mean, save_rstd, scratch, dx, dw, db,
                     Nb, C, HW, training, (hipStream_t)s);
}
void launch_bn_bwd_bf16(const unsigned short* x, const unsigned short* dy,
                        const float* w, const float* save_mean,
                        const float* save_rstd, float* scratch,
                        unsigned short* dx, float* dw, float* db, int Nb,
                        int C, int HW, int training, void* s) {
  bn_bwd_impl<unsigned short>(x, dy, w, save_mean, save_rstd, scratch, dx,
                              dw, db, Nb, C, HW, training, (hipStream_t)s);
}
}
