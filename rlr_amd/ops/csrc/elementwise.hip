// Elementwise / pooling / dropout / CE-loss kernels (gfx950).
// Covers SURVEY.md §2b K3 (relu), K4 (maxpool2x2), K5 (dropout, philox),
// K7 (cross-entropy), K18 (eval reductions) plus the build's add_relu and
// global-avg-pool for ResNet18.
// All memory-bound: float4-vectorized grid-stride loops at the HBM roofline
// (cdna_hip_programming.md Appendix B, Guideline 13).
#include "common.h"

// ---------------------------------------------------------------- relu

__global__ void relu_fwd_k(const float* __restrict__ x, float* __restrict__ y,
                           long n) {
  long i4 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long n4 = n / 4;
  for (long i = i4; i < n4; i += stride) {
    float4 v = ((const float4*)x)[i];
    v.x = fmaxf(v.x, 0.f); v.y = fmaxf(v.y, 0.f);
    v.z = fmaxf(v.z, 0.f); v.w = fmaxf(v.w, 0.f);
    ((float4*)y)[i] = v;
  }
  for (long i = n4 * 4 + i4; i < n; i += stride) y[i] = fmaxf(x[i], 0.f);
}

// dx = dy * (y > 0): the mask comes from the POST-activation output, which
// is what the fused conv/linear+relu forwards save.
__global__ void relu_bwd_k(const float* __restrict__ y,
                           const float* __restrict__ dy,
                           float* __restrict__ dx, long n) {
  long i4 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long n4 = n / 4;
  for (long i = i4; i < n4; i += stride) {
    float4 a = ((const float4*)y)[i];
    float4 g = ((const float4*)dy)[i];
    g.x = a.x > 0.f ? g.x : 0.f; g.y = a.y > 0.f ? g.y : 0.f;
    g.z = a.z > 0.f ? g.z : 0.f; g.w = a.w > 0.f ? g.w : 0.f;
    ((float4*)dx)[i] = g;
  }
  for (long i = n4 * 4 + i4; i < n; i += stride)
    dx[i] = y[i] > 0.f ? dy[i] : 0.f;
}

__global__ void add_relu_k(const float* __restrict__ a,
                           const float* __restrict__ b,
                           float* __restrict__ y, long n) {
  long i4 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long n4 = n / 4;
  for (long i = i4; i < n4; i += stride) {
    float4 va = ((const float4*)a)[i];
    float4 vb = ((const float4*)b)[i];
    va.x = fmaxf(va.x + vb.x, 0.f); va.y = fmaxf(va.y + vb.y, 0.f);
    va.z = fmaxf(va.z + vb.z, 0.f); va.w = fmaxf(va.w + vb.w, 0.f);
    ((float4*)y)[i] = va;
  }
  for (long i = n4 * 4 + i4; i < n; i += stride)
    y[i] = fmaxf(a[i] + b[i], 0.f);
}

extern "C" {
void launch_relu_fwd(const float* x, float* y, long n, void* s) {
  relu_fwd_k<<<grid_for(n / 4 + 1), kBlock, 0, (hipStream_t)s>>>(x, y, n);
}
void launch_relu_bwd(const float* y, const float* dy, float* dx, long n,
                     void* s) {
  relu_bwd_k<<<grid_for(n / 4 + 1), kBlock, 0, (hipStream_t)s>>>(y, dy, dx, n);
}
void launch_add_relu(const float* a, const float* b, float* y, long n,
                     void* s) {
  add_relu_k<<<grid_for(n / 4 + 1), kBlock, 0, (hipStream_t)s>>>(a, b, y, n);
}
}

// in-place a += b (residual gradient join in the manual tape)
__global__ void add_inplace_k(float* __restrict__ a,
                              const float* __restrict__ b, long n) {
  long i4 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long n4 = n / 4;
  for (long i = i4; i < n4; i += stride) {
    float4 va = ((const float4*)a)[i];
    float4 vb = ((const float4*)b)[i];
    va.x += vb.x; va.y += vb.y; va.z += vb.z; va.w += vb.w;
    ((float4*)a)[i] = va;
  }
  for (long i = n4 * 4 + i4; i < n; i += stride) a[i] += b[i];
}

typedef unsigned short ush8 __attribute__((ext_vector_type(8)));

__global__ void add_inplace_bf16_k(unsigned short* __restrict__ a,
                                   const unsigned short* __restrict__ b,
                                   long n) {
  long i1 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long n8 = n / 8;
  for (long i = i1; i < n8; i += stride) {
    ush8 va = ((const ush8*)a)[i];
    ush8 vb = ((const ush8*)b)[i];
    ush8 out;
#pragma unroll
    for (int e = 0; e < 8; ++e)
      out[e] = f2bf_(bf2f_(va[e]) + bf2f_(vb[e]));
    ((ush8*)a)[i] = out;
  }
  for (long i = n8 * 8 + i1; i < n; i += stride)
    a[i] = f2bf_(bf2f_(a[i]) + bf2f_(b[i]));
}

// masked in-place join: a = (y>0) ? a+b : 0 — the residual gradient join
// fused with the upstream relu mask (manual tape; bitwise-equal to add
// followed by relu_bwd since masking commutes with the add)
__global__ void add_relu_bwd_bf16_k(unsigned short* __restrict__ a,
                                    const unsigned short* __restrict__ b,
                                    const unsigned short* __restrict__ y,
                                    long n) {
  long i1 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long n8 = n / 8;
  for (long i = i1; i < n8; i += stride) {
    ush8 va = ((const ush8*)a)[i];
    ush8 vb = ((const ush8*)b)[i];
    ush8 vy = ((const ush8*)y)[i];
    ush8 out;
#pragma unroll
    for (int e = 0; e < 8; ++e)
      out[e] = bf2f_(vy[e]) > 0.f
                   ? f2bf_(bf2f_(va[e]) + bf2f_(vb[e]))
                   : (unsigned short)0;
    ((ush8*)a)[i] = out;
  }
  for (long i = n8 * 8 + i1; i < n; i += stride)
    a[i] = bf2f_(y[i]) > 0.f ? f2bf_(bf2f_(a[i]) + bf2f_(b[i]))
                             : (unsigned short)0;
}

__global__ void add_relu_bwd_f32_k(float* __restrict__ a,
                                   const float* __restrict__ b,
                                   const float* __restrict__ y, long n) {
  long i4 = (long)blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  long n4 = n / 4;
  for (long i = i4; i < n4; i += stride) {
    float4 va = ((const float4*)a)[i];
    float4 vb = ((const float4*)b)[i];
    float4 vy = ((const float4*)y)[i];
    va.x = vy.x > 0.f ? va.x + vb.x : 0.f;
    va.y = vy.y > 0.f ? va.y + vb.y : 0.f;
    va.z = vy.z > 0.f ? va.z + vb.z : 0.f;
    va.w = vy.w > 0.f ? va.w + vb.w : 0.f;
    ((float4*)a)[i] = va;
  }
  for (long i = n4 * 4 + i4; i < n; i += stride)
    a[i] = y[i] > 0.f ? a[i] + b[i] : 0.f;
}

extern "C" {
void launch_add_relu_bwd(float* a, const float* b, const float* y, long n,
                         void* s) {
  add_relu_bwd_f32_k<<<grid_for(n / 4 + 1), kBlock, 0, (hipStream_t)s>>>(
      a, b, y, n);
}
void launch_add_relu_bwd_bf16(unsigned short* a, const unsigned short* b,
                              const unsigned short* y, long n, void* s) {
  add_relu_bwd_bf16_k<<<grid_for(n / 8 + 1), kBlock, 0, (hipStream_t)s>>>(
      a, b, y, n);
}
void launch_add_inplace(float* a, const float* b, long n, void* s) {
  add_inplace_k<<<grid_for(n / 4 + 1), kBlock, 0, (hipStream_t)s>>>(a, b, n);
}
void launch_add_inplace_bf16(unsigned short* a, const unsigned short* b,
                             long n, void* s) {
  add_inplace_bf16_k<<<grid_for(n / 8 + 1), kBlock, 0, (hipStream_t)s>>>(
      a, b, n);
}
}

// ------------------------------------------------------- maxpool 2x2 NHWC

// channels_last: c is the fastest axis, so threads over the flat index are
// coalesced.  idx stores the 2x2 argmax (0..3) for the gather backward.
template <typename T>
__global__ void maxpool2x2_fwd_k(const T* __restrict__ x,
                                 T* __restrict__ y,
                                 uint8_t* __restrict__ idx,
                                 long B, int H, int W, int OH, int OW,
                                 int C) {
  long n_out = B * OH * (long)OW * C;
  long stride = (long)gridDim.x * blockDim.x;
  int WC = W * C;
  // C is a power of two in every model here: shift/mask decomposition
  int csh = 31 - __clz(C);
  bool cp2 = (C & (C - 1)) == 0;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n_out;
       i += stride) {
    int c = cp2 ? (int)(i & (C - 1)) : (int)(i % C);
    long rest = cp2 ? (i >> csh) : (i / C);
    int ow = rest % OW;
    int oh = (rest / OW) % OH;
    long b = rest / ((long)OW * OH);
    const T* p = x + ((b * H + 2 * oh) * (long)W + 2 * ow) * C + c;
    float v00 = ldv(p), v01 = ldv(p + C), v10 = ldv(p + WC),
          v11 = ldv(p + WC + C);
    float m = v00; uint8_t a = 0;
    if (v01 > m) { m = v01; a = 1; }
    if (v10 > m) { m = v10; a = 2; }
    if (v11 > m) { m = v11; a = 3; }
    stv(&y[i], m);
    idx[i] = a;
  }
}

// Gather form: one thread per INPUT element (write-once, no zero-init);
// the default (RLR_MP_STRIP=1 selects the strip form below).
template <typename T>
__global__ void maxpool2x2_bwd_gather_k(const T* __restrict__ dy,
                                        const uint8_t* __restrict__ idx,
                                        T* __restrict__ dx,
                                        long B, int H, int W, int OH,
                                        int OW, int C) {
  long n_in = B * H * (long)W * C;
  long stride = (long)gridDim.x * blockDim.x;
  int csh = 31 - __clz(C);
  bool cp2 = (C & (C - 1)) == 0;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n_in;
       i += stride) {
    int c = cp2 ? (int)(i & (C - 1)) : (int)(i % C);
    long rest = cp2 ? (i >> csh) : (i / C);
    int iw = rest % W;
    int ih = (rest / W) % H;
    long b = rest / ((long)W * H);
    int oh = ih >> 1, ow = iw >> 1;
    float g = 0.f;
    if (oh < OH && ow < OW) {
      long o = ((b * OH + oh) * (long)OW + ow) * C + c;
      uint8_t a = ((ih & 1) << 1) | (iw & 1);
      if (idx[o] == a) g = ldv(&dy[o]);
    }
    stv(&dx[i], g);
  }
}

// float4 gather (C % 4 == 0 — every pooled activation here): 4 channels
// per thread share one (ih,iw)->(oh,ow) decomposition; idx is a uchar4,
// dy/dx are float4 — a quarter of the address math and store issues of
// the scalar form.  `relu_pooled` (optional) fuses the upstream relu
// mask: for post-relu inputs the window max equals the pooled value, so
// masking dy by pooled>0 is bitwise-identical to a separate relu_bwd on
// the pre-pool tensor (manual tape, models/cnn.py).
__global__ void maxpool2x2_bwd_gather4_k(const float* __restrict__ dy,
                                         const uint8_t* __restrict__ idx,
                                         const float* __restrict__ relu_pooled,
                                         float* __restrict__ dx, long B,
                                         int H, int W, int OH, int OW,
                                         int C) {
  long n4 = B * H * (long)W * C / 4;
  long stride = (long)gridDim.x * blockDim.x;
  int c4 = C / 4;
  int csh = 31 - __clz(c4);
  bool cp2 = (c4 & (c4 - 1)) == 0;
  for (long i4 = (long)blockIdx.x * blockDim.x + threadIdx.x; i4 < n4;
       i4 += stride) {
    int c = (cp2 ? (int)(i4 & (c4 - 1)) : (int)(i4 % c4)) * 4;
    long rest = cp2 ? (i4 >> csh) : (i4 / c4);
    int iw = rest % W;
    int ih = (rest / W) % H;
    long b = rest / ((long)W * H);
    int oh = ih >> 1, ow = iw >> 1;
    float4 g = {0.f, 0.f, 0.f, 0.f};
    if (oh < OH && ow < OW) {
      long o = ((b * OH + oh) * (long)OW + ow) * C + c;
      const uint8_t* ap = idx + o;
      float4 d4 = *(const float4*)(dy + o);
      uint8_t a = ((ih & 1) << 1) | (iw & 1);
      g.x = ap[0] == a ? d4.x : 0.f;
      g.y = ap[1] == a ? d4.y : 0.f;
      g.z = ap[2] == a ? d4.z : 0.f;
      g.w = ap[3] == a ? d4.w : 0.f;
      if (relu_pooled) {
        float4 p4 = *(const float4*)(relu_pooled + o);
        g.x = p4.x > 0.f ? g.x : 0.f;
        g.y = p4.y > 0.f ? g.y : 0.f;
        g.z = p4.z > 0.f ? g.z : 0.f;
        g.w = p4.w > 0.f ? g.w : 0.f;
      }
    }
    *(float4*)(dx + i4 * 4) = g;
  }
}

// Strip form: one thread per (b, oh, c) output ROW STRIP.  Walking ow in
// a loop removes the per-element div/mod of the earlier gather form (the
// only VALU divisions on this path; they held the kernel to ~1.5 TB/s
// while relu_bwd streams at ~2.8).  Every input cell belongs to exactly
// one 2x2 window (stride 2), so each is written exactly once; odd tail
// rows/cols (H or W > 2*OH/2*OW) receive zero gradient, matching the
// reference pool's floor semantics.
template <typename T>
__global__ void maxpool2x2_bwd_k(const T* __restrict__ dy,
                                 const uint8_t* __restrict__ idx,
                                 T* __restrict__ dx,
                                 long B, int H, int W, int OH, int OW,
                                 int C) {
  long n_strip = B * OH * (long)C;
  long stride = (long)gridDim.x * blockDim.x;
  int WC = W * C;
  int csh = 31 - __clz(C);
  bool cp2 = (C & (C - 1)) == 0;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n_strip;
       i += stride) {
    int c = cp2 ? (int)(i & (C - 1)) : (int)(i % C);
    long rest = cp2 ? (i >> csh) : (i / C);
    int oh = rest % OH;
    long b = rest / OH;
    const T* dyp = dy + ((b * OH + oh) * (long)OW) * C + c;
    const uint8_t* ip = idx + ((b * OH + oh) * (long)OW) * C + c;
    T* dxp = dx + ((b * H + 2 * oh) * (long)W) * C + c;
    for (int ow = 0; ow < OW; ++ow) {
      float g = ldv(dyp + (long)ow * C);
      uint8_t a = ip[(long)ow * C];
      long p = 2L * ow * C;
      stv(dxp + p, a == 0 ? g : 0.f);
      stv(dxp + p + C, a == 1 ? g : 0.f);
      stv(dxp + p + WC, a == 2 ? g : 0.f);
      stv(dxp + p + WC + C, a == 3 ? g : 0.f);
    }
    // odd input tail columns/rows get zero gradient
    for (int iw = 2 * OW; iw < W; ++iw) {
      stv(dxp + (long)iw * C, 0.f);
      stv(dxp + (long)iw * C + WC, 0.f);
    }
    if (oh == OH - 1)
      for (int ih = 2 * OH; ih < H; ++ih)
        for (int iw = 0; iw < W; ++iw)
          stv(dx + ((b * H + ih) * (long)W + iw) * C + c, 0.f);
  }
}

static bool mp_strip() {
  static int v = -1;
  if (v < 0) {
    const char* e = getenv("RLR_MP_STRIP");
    // A/B on MI355X was a tie at bench scale (4.21/4.21 gather vs
    // 4.13/4.34 strip) — keep the simpler gather form unless asked
    v = (e && e[0] == '1') ? 1 : 0;
  }
  return v == 1;
}

extern "C" {
void launch_maxpool2x2_fwd(const float* x, float* y, uint8_t* idx, long B,
                           int H, int W, int OH, int OW, int C, void* s) {
  long n = B * OH * (long)OW * C;
  maxpool2x2_fwd_k<float><<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(
      x, y, idx, B, H, W, OH, OW, C);
}
void launch_maxpool2x2_bwd(const float* dy, const uint8_t* idx, float* dx,
                           long B, int H, int W, int OH, int OW, int C,
                           void* s) {
  if ((C % 4) == 0) {
    long n4 = B * H * (long)W * C / 4;
    maxpool2x2_bwd_gather4_k<<<grid_for(n4), kBlock, 0, (hipStream_t)s>>>(
        dy, idx, nullptr, dx, B, H, W, OH, OW, C);
    return;
  }
  if (mp_strip()) {
    long n = B * OH * (long)C;  // one thread per (b, oh, c) strip
    maxpool2x2_bwd_k<float><<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(
        dy, idx, dx, B, H, W, OH, OW, C);
  } else {
    long n = B * H * (long)W * C;
    maxpool2x2_bwd_gather_k<float>
        <<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(dy, idx, dx, B, H, W,
                                                     OH, OW, C);
  }
}

// fused relu-mask variant (C % 4 == 0 required; manual tape only)
void launch_maxpool2x2_bwd_relu(const float* dy, const uint8_t* idx,
                                const float* relu_pooled, float* dx, long B,
                                int H, int W, int OH, int OW, int C,
                                void* s) {
  long n4 = B * H * (long)W * C / 4;
  maxpool2x2_bwd_gather4_k<<<grid_for(n4), kBlock, 0, (hipStream_t)s>>>(
      dy, idx, relu_pooled, dx, B, H, W, OH, OW, C);
}
void launch_maxpool2x2_fwd_bf16(const unsigned short* x, unsigned short* y,
                                uint8_t* idx, long B, int H, int W, int OH,
                                int OW, int C, void* s) {
  long n = B * OH * (long)OW * C;
  maxpool2x2_fwd_k<unsigned short>
      <<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(x, y, idx, B, H, W, OH,
                                                   OW, C);
}
void launch_maxpool2x2_bwd_bf16(const unsigned short* dy,
                                const uint8_t* idx, unsigned short* dx,
                                long B, int H, int W, int OH, int OW, int C,
                                void* s) {
  if (!mp_strip()) {
    long n2 = B * H * (long)W * C;
    maxpool2x2_bwd_gather_k<unsigned short>
        <<<grid_for(n2), kBlock, 0, (hipStream_t)s>>>(dy, idx, dx, B, H, W,
                                                      OH, OW, C);
    return;
  }
  long n = B * OH * (long)C;  // one thread per (b, oh, c) strip
  maxpool2x2_bwd_k<unsigned short>
      <<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(dy, idx, dx, B, H, W, OH,
                                                   OW, C);
}
}

// ----------------------------------------------------- global avgpool NHWC

// out[b][c] = mean over HW; thread per (b,c): consecutive threads read
// consecutive c (coalesced column walk).
__global__ void gap_fwd_k(const float* __restrict__ x, float* __restrict__ y,
                          long B, int HW, int C) {
  long n = B * C;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int c = i % C;
    long b = i / C;
    const float* p = x + b * (long)HW * C + c;
    float acc = 0.f;
    for (int hw = 0; hw < HW; ++hw) acc += p[(long)hw * C];
    y[i] = acc / HW;
  }
}

__global__ void gap_bwd_k(const float* __restrict__ dy,
                          float* __restrict__ dx, long B, int HW, int C) {
  long n = B * (long)HW * C;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int c = i % C;
    long b = i / ((long)HW * C);
    dx[i] = dy[b * C + c] / HW;
  }
}

extern "C" {
void launch_gap_fwd(const float* x, float* y, long B, int HW, int C,
                    void* s) {
  gap_fwd_k<<<grid_for(B * C), kBlock, 0, (hipStream_t)s>>>(x, y, B, HW, C);
}
void launch_gap_bwd(const float* dy, float* dx, long B, int HW, int C,
                    void* s) {
  gap_bwd_k<<<grid_for(B * (long)HW * C), kBlock, 0, (hipStream_t)s>>>(
      dy, dx, B, HW, C);
}
}

// ---------------------------------------------------------------- dropout

// Elementwise Bernoulli(1-p) * 1/(1-p); 4 elements per philox draw.
__global__ void dropout_fwd_k(const float* __restrict__ x,
                              float* __restrict__ y,
                              uint8_t* __restrict__ mask, long n, float p,
                              uint64_t seed, uint64_t offset) {
  float scale = 1.0f / (1.0f - p);
  long stride = (long)gridDim.x * blockDim.x;
  long n4 = (n + 3) / 4;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    Philox4 r = philox4(seed, offset, (uint32_t)i);
    uint32_t rv[4] = {r.x, r.y, r.z, r.w};
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      long k = i * 4 + j;
      if (k < n) {
        bool keep = u32_to_uniform(rv[j]) >= p;
        mask[k] = keep;
        y[k] = keep ? x[k] * scale : 0.f;
      }
    }
  }
}

// Device-state variant for hipGraph capture: seed/base-offset live in a
// 2 x u64 device buffer bumped between replays, so masks advance across
// graph replays without re-capture.
__global__ void dropout_fwd_dev_k(const float* __restrict__ x,
                                  float* __restrict__ y,
                                  uint8_t* __restrict__ mask, long n,
                                  float p,
                                  const unsigned long long* __restrict__
                                      state, int site) {
  uint64_t seed = state[0];
  uint64_t offset = state[1] + site;
  float scale = 1.0f / (1.0f - p);
  long stride = (long)gridDim.x * blockDim.x;
  long n4 = (n + 3) / 4;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    Philox4 r = philox4(seed, offset, (uint32_t)i);
    uint32_t rv[4] = {r.x, r.y, r.z, r.w};
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      long k = i * 4 + j;
      if (k < n) {
        bool keep = u32_to_uniform(rv[j]) >= p;
        mask[k] = keep;
        y[k] = keep ? x[k] * scale : 0.f;
      }
    }
  }
}

__global__ void dropout_bwd_k(const float* __restrict__ dy,
                              const uint8_t* __restrict__ mask,
                              float* __restrict__ dx, long n, float p) {
  float scale = 1.0f / (1.0f - p);
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    dx[i] = mask[i] ? dy[i] * scale : 0.f;
}

// fused dropout+relu backward (manual tape): out = (y>0) * dropout_bwd —
// bitwise-identical to dropout_bwd_k followed by relu_bwd_k
__global__ void dropout_relu_bwd_k(const float* __restrict__ dy,
                                   const uint8_t* __restrict__ mask,
                                   const float* __restrict__ y,
                                   float* __restrict__ dx, long n, float p) {
  float scale = 1.0f / (1.0f - p);
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    dx[i] = (y[i] > 0.f && mask[i]) ? dy[i] * scale : 0.f;
}

extern "C" {
void launch_dropout_fwd(const float* x, float* y, uint8_t* mask, long n,
                        float p, uint64_t seed, uint64_t offset, void* s) {
  dropout_fwd_k<<<grid_for((n + 3) / 4), kBlock, 0, (hipStream_t)s>>>(
      x, y, mask, n, p, seed, offset);
}
void launch_dropout_fwd_dev(const float* x, float* y, uint8_t* mask, long n,
                            float p, const unsigned long long* state,
                            int site, void* s) {
  dropout_fwd_dev_k<<<grid_for((n + 3) / 4), kBlock, 0, (hipStream_t)s>>>(
      x, y, mask, n, p, state, site);
}
void launch_dropout_bwd(const float* dy, const uint8_t* mask, float* dx,
                        long n, float p, void* s) {
  dropout_bwd_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(dy, mask, dx, n,
                                                            p);
}
void launch_dropout_relu_bwd(const float* dy, const uint8_t* mask,
                             const float* y, float* dx, long n, float p,
                             void* s) {
  dropout_relu_bwd_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(
      dy, mask, y, dx, n, p);
}
}

// ------------------------------------------------------------ cross entropy

// Mean-reduced CE over (B, C<=64) logits.  One wave per row: lane c holds
// logit c; wave shuffle reduce for max and sum.  Saves softmax for bwd.
__global__ void ce_fwd_k(const float* __restrict__ logits,
                         const long* __restrict__ labels,
                         float* __restrict__ softmax,
                         float* __restrict__ row_loss, int B, int C) {
  int row = blockIdx.x * (blockDim.x / kWave) + threadIdx.x / kWave;
  int lane = threadIdx.x % kWave;
  if (row >= B) return;
  float v = (lane < C) ? logits[(long)row * C + lane] : -3.4e38f;
  float m = v;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    m = fmaxf(m, __shfl_xor(m, off, kWave));
  float e = (lane < C) ? __expf(v - m) : 0.f;
  float sum = e;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    sum += __shfl_xor(sum, off, kWave);
  float sm = e / sum;
  if (lane < C) softmax[(long)row * C + lane] = sm;
  if (lane == 0) {
    long t = labels[row];
    float lt = logits[(long)row * C + t];
    row_loss[row] = logf(sum) + m - lt;
  }
}

// single-block deterministic mean over row losses
__global__ void ce_reduce_k(const float* __restrict__ row_loss,
                            float* __restrict__ out, int B) {
  __shared__ float sh[kBlock];
  float acc = 0.f;
  for (int i = threadIdx.x; i < B; i += blockDim.x) acc += row_loss[i];
  sh[threadIdx.x] = acc;
  __syncthreads();
  for (int off = kBlock / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) sh[threadIdx.x] += sh[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) out[0] = sh[0] / B;
}

// dlogits = dloss * (softmax - onehot) / B
__global__ void ce_bwd_k(const float* __restrict__ softmax,
                         const long* __restrict__ labels,
                         const float* __restrict__ dloss,
                         float* __restrict__ dlogits, int B, int C) {
  long n = (long)B * C;
  long stride = (long)gridDim.x * blockDim.x;
  float g = dloss[0];
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int c = i % C;
    long row = i / C;
    float v = softmax[i] - (labels[row] == c ? 1.f : 0.f);
    dlogits[i] = g * v / B;
  }
}

extern "C" {
void launch_ce_fwd(const float* logits, const long* labels, float* softmax,
                   float* row_loss, float* loss_out, int B, int C, void* s) {
  int wpb = kBlock / kWave;
  ce_fwd_k<<<(B + wpb - 1) / wpb, kBlock, 0, (hipStream_t)s>>>(
      logits, labels, softmax, row_loss, B, C);
  ce_reduce_k<<<1, kBlock, 0, (hipStream_t)s>>>(row_loss, loss_out, B);
}
void launch_ce_bwd(const float* softmax, const long* labels,
                   const float* dloss, float* dlogits, int B, int C,
                   void* s) {
  ce_bwd_k<<<grid_for((long)B * C), kBlock, 0, (hipStream_t)s>>>(
      softmax, labels, dloss, dlogits, B, C);
}
}

// ------------------------------------------------------------- eval fused

// Fused eval reductions: per-row CE loss + argmax + confusion scatter.
// Confusion counts are integer-valued float atomicAdds (exact, order-free);
// the loss sum is reduced single-block for determinism.
__global__ void eval_update_k(const float* __restrict__ logits,
                              const long* __restrict__ labels,
                              float* __restrict__ conf,
                              double* __restrict__ loss_sum, int B, int C,
                              int num_classes) {
  __shared__ double sh[kBlock];
  double acc = 0.0;
  for (int row = threadIdx.x; row < B; row += blockDim.x) {
    const float* l = logits + (long)row * C;
    float m = l[0];
    int arg = 0;
    for (int c = 1; c < C; ++c)
      if (l[c] > m) { m = l[c]; arg = c; }
    float sum = 0.f;
    for (int c = 0; c < C; ++c) sum += __expf(l[c] - m);
    long t = labels[row];
    acc += (double)(logf(sum) + m - l[t]);
    atomicAdd(&conf[t * num_classes + arg], 1.0f);
  }
  sh[threadIdx.x] = acc;
  __syncthreads();
  for (int off = kBlock / 2; off > 0; off >>= 1) {
    if (threadIdx.x < off) sh[threadIdx.x] += sh[threadIdx.x + off];
    __syncthreads();
  }
  if (threadIdx.x == 0) loss_sum[0] += sh[0];
}

extern "C" {
void launch_eval_update(const float* logits, const long* labels, float* conf,
                        double* loss_sum, int B, int C, int num_classes,
                        void* s) {
  eval_update_k<<<1, kBlock, 0, (hipStream_t)s>>>(logits, labels, conf,
                                                  loss_sum, B, C,
                                                  num_classes);
}
}

// -------------------------------------------------- bf16 scalar variants

__global__ void relu_fwd_bf16_k(const unsigned short* __restrict__ x,
                                unsigned short* __restrict__ y, long n) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    y[i] = bf2f_(x[i]) > 0.f ? x[i] : (unsigned short)0;
}

__global__ void relu_bwd_bf16_k(const unsigned short* __restrict__ y,
                                const unsigned short* __restrict__ dy,
                                unsigned short* __restrict__ dx, long n) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    dx[i] = bf2f_(y[i]) > 0.f ? dy[i] : (unsigned short)0;
}

__global__ void add_relu_bf16_k(const unsigned short* __restrict__ a,
                                const unsigned short* __restrict__ b,
                                unsigned short* __restrict__ y, long n) {
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    y[i] = f2bf_(fmaxf(bf2f_(a[i]) + bf2f_(b[i]), 0.f));
}

__global__ void dropout_fwd_dev_bf16_k(const unsigned short* __restrict__ x,
                                       unsigned short* __restrict__ y,
                                       uint8_t* __restrict__ mask, long n,
                                       float p,
                                       const unsigned long long* __restrict__
                                           state, int site) {
  uint64_t seed = state[0];
  uint64_t offset = state[1] + site;
  float scale = 1.0f / (1.0f - p);
  long stride = (long)gridDim.x * blockDim.x;
  long n4 = (n + 3) / 4;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n4;
       i += stride) {
    Philox4 r = philox4(seed, offset, (uint32_t)i);
    uint32_t rv[4] = {r.x, r.y, r.z, r.w};
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      long k = i * 4 + j;
      if (k < n) {
        bool keep = u32_to_uniform(rv[j]) >= p;
        mask[k] = keep;
        y[k] = keep ? f2bf_(bf2f_(x[k]) * scale) : (unsigned short)0;
      }
    }
  }
}

__global__ void dropout_bwd_bf16_k(const unsigned short* __restrict__ dy,
                                   const uint8_t* __restrict__ mask,
                                   unsigned short* __restrict__ dx, long n,
                                   float p) {
  float scale = 1.0f / (1.0f - p);
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride)
    dx[i] = mask[i] ? f2bf_(bf2f_(dy[i]) * scale) : (unsigned short)0;
}

__global__ void gap_fwd_bf16_k(const unsigned short* __restrict__ x,
                               unsigned short* __restrict__ y, long B,
                               int HW, int C) {
  long n = B * C;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int c = i % C;
    long b = i / C;
    const unsigned short* p = x + b * (long)HW * C + c;
    float acc = 0.f;
    for (int hw = 0; hw < HW; ++hw) acc += bf2f_(p[(long)hw * C]);
    y[i] = f2bf_(acc / HW);
  }
}

__global__ void gap_bwd_bf16_k(const unsigned short* __restrict__ dy,
                               unsigned short* __restrict__ dx, long B,
                               int HW, int C) {
  long n = B * (long)HW * C;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int c = i % C;
    long b = i / ((long)HW * C);
    dx[i] = f2bf_(bf2f_(dy[b * C + c]) / HW);
  }
}

extern "C" {
void launch_relu_fwd_bf16(const unsigned short* x, unsigned short* y,
                          long n, void* s) {
  relu_fwd_bf16_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(x, y, n);
}
void launch_relu_bwd_bf16(const unsigned short* y, const unsigned short* dy,
                          unsigned short* dx, long n, void* s) {
  relu_bwd_bf16_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(y, dy, dx, n);
}
void launch_add_relu_bf16(const unsigned short* a, const unsigned short* b,
                          unsigned short* y, long n, void* s) {
  add_relu_bf16_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(a, b, y, n);
}
void launch_dropout_fwd_dev_bf16(const unsigned short* x, unsigned short* y,
                                 uint8_t* mask, long n, float p,
                                 const unsigned long long* state, int site,
                                 void* s) {
  dropout_fwd_dev_bf16_k<<<grid_for((n + 3) / 4), kBlock, 0,
                           (hipStream_t)s>>>(x, y, mask, n, p, state, site);
}
void launch_dropout_bwd_bf16(const unsigned short* dy, const uint8_t* mask,
                             unsigned short* dx, long n, float p, void* s) {
  dropout_bwd_bf16_k<<<grid_for(n), kBlock, 0, (hipStream_t)s>>>(dy, mask,
                                                                 dx, n, p);
}
void launch_gap_fwd_bf16(const unsigned short* x, unsigned short* y, long B,
                         int HW, int C, void* s) {
  gap_fwd_bf16_k<<<grid_for(B * C), kBlock, 0, (hipStream_t)s>>>(x, y, B,
                                                                 HW, C);
}
void launch_gap_bwd_bf16(const unsigned short* dy, unsigned short* dx,
                         long B, int HW, int C, void* s) {
  gap_bwd_bf16_k<<<grid_for(B * (long)HW * C), kBlock, 0,
                   (hipStream_t)s>>>(dy, dx, B, HW, C);
}
}

// ---------------------------------------------- NHWC <-> flat CHW permute
// The fc layers consume the conv trunk's channels_last output flattened in
// the reference's CHW order (models.py:21,50 parity).  Torch's generic
// permute kernel measured ~12.5 us for this shape; these are ~roofline.

// LDS-tiled per-image transpose: in[b][r][c] -> out[b][c][r], both sides
// coalesced (the naive flatten/unflatten kernels were strided on one side:
// consecutive lanes 256 B apart -> 16x L2 over-fetch; measured 15.6/14.1 us
// for a 2.4 MB tensor).  32x32 tiles, 256 threads (32x8), +1 pad column.
template <typename T>
__global__ void batch_transpose_k(const T* __restrict__ in,
                                  T* __restrict__ out, int R, int Ccols) {
  __shared__ T tile[32][33];
  int r0 = blockIdx.x * 32, c0 = blockIdx.y * 32;
  long b = blockIdx.z;
  const T* src = in + b * (long)R * Ccols;
  T* dst = out + b * (long)R * Ccols;
  int tx = threadIdx.x & 31, ty = threadIdx.x >> 5;
#pragma unroll
  for (int i = ty; i < 32; i += 8) {
    int r = r0 + i, c = c0 + tx;
    if (r < R && c < Ccols) tile[i][tx] = src[(long)r * Ccols + c];
  }
  __syncthreads();
#pragma unroll
  for (int i = ty; i < 32; i += 8) {
    int c = c0 + i, r = r0 + tx;
    if (c < Ccols && r < R) dst[(long)c * R + r] = tile[tx][i];
  }
}

// (the scalar flatten/unflatten kernels were replaced by the
// LDS-tiled batch_transpose_k above)

extern "C" {
void launch_nhwc_flatten(const float* in, float* out, long B, int C, int H,
                         int W, void* s) {
  int HW = H * W;
  dim3 g((HW + 31) / 32, (C + 31) / 32, B);
  batch_transpose_k<float><<<g, 256, 0, (hipStream_t)s>>>(in, out, HW, C);
}
void launch_nhwc_unflatten(const float* in, float* out, long B, int C,
                           int H, int W, void* s) {
  int HW = H * W;
  dim3 g((C + 31) / 32, (HW + 31) / 32, B);
  batch_transpose_k<float><<<g, 256, 0, (hipStream_t)s>>>(in, out, C, HW);
}
void launch_nhwc_flatten_bf16(const unsigned short* in, unsigned short* out,
                              long B, int C, int H, int W, void* s) {
  int HW = H * W;
  dim3 g((HW + 31) / 32, (C + 31) / 32, B);
  batch_transpose_k<unsigned short><<<g, 256, 0, (hipStream_t)s>>>(in, out,
                                                                   HW, C);
}
void launch_nhwc_unflatten_bf16(const unsigned short* in,
                                unsigned short* out, long B, int C, int H,
                                int W, void* s) {
  int HW = H * W;
  dim3 g((C + 31) / 32, (HW + 31) / 32, B);
  batch_transpose_k<unsigned short><<<g, 256, 0, (hipStream_t)s>>>(in, out,
                                                                   C, HW);
}
}

// masked gap backward (manual tape): dx = (y>0) ? dy[b,c]/HW : 0 — the
// last residual block's add_relu backward folded into the gap gradient
__global__ void gap_bwd_relu_k(const float* __restrict__ dy,
                               const float* __restrict__ y,
                               float* __restrict__ dx, long B, int HW,
                               int C) {
  long n = B * (long)HW * C;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int c = i % C;
    long b = i / ((long)HW * C);
    dx[i] = y[i] > 0.f ? dy[b * C + c] / HW : 0.f;
  }
}

__global__ void gap_bwd_relu_bf16_k(const unsigned short* __restrict__ dy,
                                    const unsigned short* __restrict__ y,
                                    unsigned short* __restrict__ dx, long B,
                                    int HW, int C) {
  long n = B * (long)HW * C;
  long stride = (long)gridDim.x * blockDim.x;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += stride) {
    int c = i % C;
    long b = i / ((long)HW * C);
    dx[i] = bf2f_(y[i]) > 0.f ? f2bf_(bf2f_(dy[b * C + c]) / HW)
                              : (unsigned short)0;
  }
}

extern "C" {
void launch_gap_bwd_relu(const float* dy, const float* y, float* dx, long B,
                         int HW, int C, void* s) {
  gap_bwd_relu_k<<<grid_for(B * (long)HW * C), kBlock, 0,
                   (hipStream_t)s>>>(dy, y, dx, B, HW, C);
}
void launch_gap_bwd_relu_bf16(const unsigned short* dy,
                              const unsigned short* y, unsigned short* dx,
                              long B, int HW, int C, void* s) {
  gap_bwd_relu_bf16_k<<<grid_for(B * (long)HW * C), kBlock, 0,
                        (hipStream_t)s>>>(dy, y, dx, B, HW, C);
}
}
