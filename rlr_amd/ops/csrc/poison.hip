// On-device trojan injection + normalization (SURVEY.md §2b K19;
// reference utils.py:160-284 runs on CPU with numpy/OpenCV).
// Raw uint8 dataset storage lives in HBM; patterns are written in place
// from a coordinate table (data/poison.py pattern_spec) and batches are
// normalized on device — poisoned/clean batches never leave HBM
// (BASELINE.json requirement).
#include "common.h"

// ---- set coords: raw (B,H,W) u8 | (B,H,W,C) u8 | (B,1,H,W) f32 ----
// idxs: which images of the full dataset to poison.
__global__ void poison_set_u8_k(uint8_t* __restrict__ data,
                                const long* __restrict__ idxs, int B,
                                const int* __restrict__ coords, int P,
                                int H, int W, int C, uint8_t value) {
  long n = (long)B * P * C;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < n;
       t += stride) {
    int ch = t % C;
    int p = (t / C) % P;
    long b = idxs[t / ((long)C * P)];
    int r = coords[2 * p], c = coords[2 * p + 1];
    data[((b * H + r) * W + c) * C + ch] = value;
  }
}

__global__ void poison_set_f32_k(float* __restrict__ data,
                                 const long* __restrict__ idxs, int B,
                                 const int* __restrict__ coords, int P,
                                 int H, int W, float value) {
  long n = (long)B * P;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < n;
       t += stride) {
    int p = t % P;
    long b = idxs[t / P];
    int r = coords[2 * p], c = coords[2 * p + 1];
    data[(b * H + r) * W + c] = value;
  }
}

// ---- full-image watermark: u8 wrap-add (reference numpy uint8 +) ----
__global__ void poison_addwrap_u8_k(uint8_t* __restrict__ data,
                                    const long* __restrict__ idxs, int B,
                                    const uint8_t* __restrict__ mask, int HW) {
  long n = (long)B * HW;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < n;
       t += stride) {
    int px = t % HW;
    long b = idxs[t / HW];
    data[b * HW + px] = (uint8_t)(data[b * HW + px] + mask[px]);
  }
}

// ---- float subtract mask/255 (fedemnist watermark) ----
__global__ void poison_subf_k(float* __restrict__ data,
                              const long* __restrict__ idxs, int B,
                              const uint8_t* __restrict__ mask, int HW) {
  long n = (long)B * HW;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < n;
       t += stride) {
    int px = t % HW;
    long b = idxs[t / HW];
    data[b * HW + px] -= mask[px] / 255.0f;  // exact div: matches the CPU
                                             // path bit-for-bit
  }
}

// ---- normalize: u8 HWC raw -> f32 channels_last (NHWC storage): the
// layouts MATCH, so this is a pure elementwise pass ----
__global__ void normalize_u8_k(const uint8_t* __restrict__ raw,
                               float* __restrict__ out, long B, int H, int W,
                               int C, const float* __restrict__ mean,
                               const float* __restrict__ stdv) {
  long n = B * (long)H * W * C;
  long stride = (long)gridDim.x * blockDim.x;
  for (long t = (long)blockIdx.x * blockDim.x + threadIdx.x; t < n;
       t += stride) {
    int c = t % C;
    out[t] = (raw[t] * (1.0f / 255.0f) - mean[c]) / stdv[c];
  }
}

extern "C" {
void launch_poison_set_u8(uint8_t* data, const long* idxs, int B,
                          const int* coords, int P, int H, int W, int C,
                          int value, void* s) {
  poison_set_u8_k<<<grid_for((long)B * P * C), kBlock, 0, (hipStream_t)s>>>(
      data, idxs, B, coords, P, H, W, C, (uint8_t)value);
}
void launch_poison_set_f32(float* data, const long* idxs, int B,
                           const int* coords, int P, int H, int W,
                           float value, void* s) {
  poison_set_f32_k<<<grid_for((long)B * P), kBlock, 0, (hipStream_t)s>>>(
      data, idxs, B, coords, P, H, W, value);
}
void launch_poison_addwrap_u8(uint8_t* data, const long* idxs, int B,
                              const uint8_t* mask, int HW, void* s) {
  poison_addwrap_u8_k<<<grid_for((long)B * HW), kBlock, 0, (hipStream_t)s>>>(
      data, idxs, B, mask, HW);
}
void launch_poison_subf(float* data, const long* idxs, int B,
                        const uint8_t* mask, int HW, void* s) {
  poison_subf_k<<<grid_for((long)B * HW), kBlock, 0, (hipStream_t)s>>>(
      data, idxs, B, mask, HW);
}
void launch_normalize_u8(const uint8_t* raw, float* out, long B, int H,
                         int W, int C, const float* mean, const float* stdv,
                         void* s) {
  normalize_u8_k<<<grid_for(B * (long)H * W * C), kBlock, 0,
                   (hipStream_t)s>>>(raw, out, B, H, W, C, mean, stdv);
}
}
