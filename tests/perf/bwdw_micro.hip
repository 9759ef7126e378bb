// On-box microbench for the bf16 conv bwd-weight kernel variants.
// Times launch_conv_bwd_weight_bf16_ex (variant 0 = 64-wide crs tile,
// variant 1 = 128-wide) over the ResNet18/CIFAR shapes at several split-K
// factors on ONE box, removing box-to-box DVFS noise from the comparison.
// Also cross-checks the two variants' outputs (identical k-order per
// element -> bitwise-equal fp32 results).
//
// Build (linked against the extension objects):
//   hipcc --offload-arch=gfx950 -O3 tests/perf/bwdw_micro.hip \
//     rlr_amd/ops/csrc/build/{conv_bf16.o,conv_f32.o,gemm_f32.o} \
//     -o tests/perf/bwdw_micro
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

extern "C" {
void launch_conv_bwd_weight_bf16_ex(const unsigned short*,
                                    const unsigned short*, float*, float*,
                                    int, int, int, int, int, int, int, int,
                                    int, int, int, int, int, void*);
int conv_bwd_weight_bf16_splitk(int, int, long);
void launch_conv_bwd_weight(const float*, const float*, float*, float*,
                            int, int, int, int, int, int, int, int, int,
                            int, int, int, void*);
int conv_bwd_weight_splitk(int, int, long);
}

#define CHK(x)                                                      \
  do {                                                              \
    hipError_t e = (x);                                             \
    if (e != hipSuccess) {                                          \
      fprintf(stderr, "HIP error %s @%d\n", hipGetErrorString(e),   \
              __LINE__);                                            \
      exit(1);                                                      \
    }                                                               \
  } while (0)

static unsigned short f2bf(float f) {
  union { float f; unsigned int i; } v{f};
  unsigned int r = v.i + 0x7FFF + ((v.i >> 16) & 1);
  return (unsigned short)(r >> 16);
}

struct Shape { int Nb, C, H, W, K, R, S, stride, pad; const char* name; };

// f32 bwd-weight SK sweep over the headline CNN shapes (fp32 path feeds
// the FMNIST/CIFAR CNN bench) — same one-box method as the bf16 sweep.
static void f32_sweep(int iters) {
  Shape shapes[] = {
      {256, 32, 26, 26, 64, 3, 3, 1, 0, "F2 fmnist 32->64"},
      {256, 64, 30, 30, 128, 3, 3, 1, 0, "C2 cifar 64->128"},
      {256, 128, 13, 13, 256, 3, 3, 1, 0, "C3 cifar 128->256"},
  };
  srand(11);
  for (const Shape& sh : shapes) {
    int OH = (sh.H - sh.R) / sh.stride + 1, OW = (sh.W - sh.S) / sh.stride + 1;
    long M = (long)sh.Nb * OH * OW;
    long nx = (long)sh.Nb * sh.H * sh.W * sh.C, ndy = M * sh.K;
    int Ncrs = sh.C * sh.R * sh.S;
    std::vector<float> hx(nx), hdy(ndy);
    for (long i = 0; i < nx; ++i) hx[i] = (rand() % 1000 - 500) / 500.f;
    for (long i = 0; i < ndy; ++i) hdy[i] = (rand() % 1000 - 500) / 500.f;
    float *dx, *ddy, *dw, *ws;
    CHK(hipMalloc(&dx, nx * 4));
    CHK(hipMalloc(&ddy, ndy * 4));
    CHK(hipMalloc(&dw, (long)sh.K * Ncrs * 4));
    CHK(hipMalloc(&ws, 800L * 1024 * 1024));
    CHK(hipMemcpy(dx, hx.data(), nx * 4, hipMemcpyHostToDevice));
    CHK(hipMemcpy(ddy, hdy.data(), ndy * 4, hipMemcpyHostToDevice));
    int skp = conv_bwd_weight_splitk(sh.K, Ncrs, M);
    hipEvent_t e0, e1;
    CHK(hipEventCreate(&e0));
    CHK(hipEventCreate(&e1));
    int sks[] = {skp, 64, 128, 192, 256};
    for (int si = 0; si < 5; ++si) {
      int SK = sks[si];
      if (si > 0 && SK == skp) continue;
      if ((long)(SK + 1) * sh.K * Ncrs * 4 > 800L * 1024 * 1024) continue;
      for (int w = 0; w < 3; ++w)
        launch_conv_bwd_weight(ddy, dx, dw, ws, SK, sh.Nb, sh.C, sh.H,
                               sh.W, sh.K, sh.R, sh.S, OH, OW, sh.stride,
                               sh.pad, 0);
      CHK(hipDeviceSynchronize());
      CHK(hipEventRecord(e0));
      for (int it = 0; it < iters; ++it)
        launch_conv_bwd_weight(ddy, dx, dw, ws, SK, sh.Nb, sh.C, sh.H,
                               sh.W, sh.K, sh.R, sh.S, OH, OW, sh.stride,
                               sh.pad, 0);
      CHK(hipEventRecord(e1));
      CHK(hipEventSynchronize(e1));
      float ms;
      CHK(hipEventElapsedTime(&ms, e0, e1));
      printf("%-20s f32 SK=%-3d %8.1f us%s\n", sh.name, SK,
             ms * 1000.f / iters, SK == skp ? "  (policy)" : "");
    }
    CHK(hipFree(dx)); CHK(hipFree(ddy)); CHK(hipFree(dw)); CHK(hipFree(ws));
    CHK(hipEventDestroy(e0)); CHK(hipEventDestroy(e1));
  }
}

int main(int argc, char** argv) {
  int iters = argc > 1 ? atoi(argv[1]) : 10;
  if (argc > 2 && argv[2][0] == 'f') { f32_sweep(iters); return 0; }
  Shape shapes[] = {
      {256, 64, 32, 32, 64, 3, 3, 1, 1, "L1 64->64 32x32"},
      {256, 128, 16, 16, 128, 3, 3, 1, 1, "L2 128->128 16x16"},
      {256, 256, 8, 8, 256, 3, 3, 1, 1, "L3 256->256 8x8"},
      {256, 512, 4, 4, 512, 3, 3, 1, 1, "L4 512->512 4x4"},
      {256, 64, 32, 32, 128, 3, 3, 2, 1, "T2 64->128 s2"},
      {256, 256, 8, 8, 512, 1, 1, 2, 0, "D4 256->512 1x1 s2"},
  };
  srand(7);
  for (const Shape& sh : shapes) {
    int OH = (sh.H + 2 * sh.pad - sh.R) / sh.stride + 1;
    int OW = (sh.W + 2 * sh.pad - sh.S) / sh.stride + 1;
    long M = (long)sh.Nb * OH * OW;
    long nx = (long)sh.Nb * sh.H * sh.W * sh.C;
    long ndy = M * sh.K;
    int Ncrs = sh.C * sh.R * sh.S;
    std::vector<unsigned short> hx(nx), hdy(ndy);
    for (long i = 0; i < nx; ++i) hx[i] = f2bf((rand() % 1000 - 500) / 500.f);
    for (long i = 0; i < ndy; ++i)
      hdy[i] = f2bf((rand() % 1000 - 500) / 500.f);
    unsigned short *dx, *ddy;
    float *dw0, *dw1, *ws;
    CHK(hipMalloc(&dx, nx * 2));
    CHK(hipMalloc(&ddy, ndy * 2));
    CHK(hipMalloc(&dw0, (long)sh.K * Ncrs * 4));
    CHK(hipMalloc(&dw1, (long)sh.K * Ncrs * 4));
    CHK(hipMalloc(&ws, 600L * 1024 * 1024));
    CHK(hipMemcpy(dx, hx.data(), nx * 2, hipMemcpyHostToDevice));
    CHK(hipMemcpy(ddy, hdy.data(), ndy * 2, hipMemcpyHostToDevice));

    int skp = conv_bwd_weight_bf16_splitk(sh.K, Ncrs, M);
    long maxc = (M + 31) / 32;
    int sks[] = {skp, 1, 2, 4, 8, 16, 32, 64, 128, 256};
    hipEvent_t e0, e1;
    CHK(hipEventCreate(&e0));
    CHK(hipEventCreate(&e1));
    for (int variant = 0; variant <= 2; ++variant) {
      for (int si = 0; si < 10; ++si) {
        int SK = sks[si];
        if (SK > maxc) continue;
        bool dup = false;
        for (int j = 1; j < si; ++j) dup |= (sks[j] == SK);
        if (si > 0 && SK == skp) dup = true;
        if (dup && si > 0) continue;
        // slab bound: SK*K*Ncrs floats must fit ws
        if ((long)(SK + 1) * sh.K * Ncrs * 4 > 600L * 1024 * 1024) continue;
        float* out = variant == 0 ? dw0 : dw1;
        for (int w = 0; w < 3; ++w)
          launch_conv_bwd_weight_bf16_ex(ddy, dx, out, ws, SK, variant,
                                         sh.Nb, sh.C, sh.H, sh.W, sh.K,
                                         sh.R, sh.S, OH, OW, sh.stride,
                                         sh.pad, 0);
        CHK(hipDeviceSynchronize());
        CHK(hipEventRecord(e0));
        for (int it = 0; it < iters; ++it)
          launch_conv_bwd_weight_bf16_ex(ddy, dx, out, ws, SK, variant,
                                         sh.Nb, sh.C, sh.H, sh.W, sh.K,
                                         sh.R, sh.S, OH, OW, sh.stride,
                                         sh.pad, 0);
        CHK(hipEventRecord(e1));
        CHK(hipEventSynchronize(e1));
        float ms;
        CHK(hipEventElapsedTime(&ms, e0, e1));
        printf("%-20s v%d SK=%-3d %8.1f us%s\n", sh.name, variant, SK,
               ms * 1000.f / iters, SK == skp ? "  (policy)" : "");
      }
    }
    // cross-check at policy SK (same k order per element -> equal)
    std::vector<float> h0((long)sh.K * Ncrs), h1((long)sh.K * Ncrs);
    launch_conv_bwd_weight_bf16_ex(ddy, dx, dw0, ws, skp, 0, sh.Nb, sh.C,
                                   sh.H, sh.W, sh.K, sh.R, sh.S, OH, OW,
                                   sh.stride, sh.pad, 0);
    launch_conv_bwd_weight_bf16_ex(ddy, dx, dw1, ws, skp, 1, sh.Nb, sh.C,
                                   sh.H, sh.W, sh.K, sh.R, sh.S, OH, OW,
                                   sh.stride, sh.pad, 0);
    CHK(hipMemcpy(h0.data(), dw0, h0.size() * 4, hipMemcpyDeviceToHost));
    CHK(hipMemcpy(h1.data(), dw1, h1.size() * 4, hipMemcpyDeviceToHost));
    double md = 0;
    for (size_t i = 0; i < h0.size(); ++i)
      md = fmax(md, fabs((double)h0[i] - h1[i]));
    printf("%-20s v0-vs-v1 max|diff| = %g %s\n", sh.name, md,
           md == 0 ? "OK" : "MISMATCH");
    launch_conv_bwd_weight_bf16_ex(ddy, dx, dw1, ws, skp, 2, sh.Nb, sh.C,
                                   sh.H, sh.W, sh.K, sh.R, sh.S, OH, OW,
                                   sh.stride, sh.pad, 0);
    CHK(hipMemcpy(h1.data(), dw1, h1.size() * 4, hipMemcpyDeviceToHost));
    md = 0;
    for (size_t i = 0; i < h0.size(); ++i)
      md = fmax(md, fabs((double)h0[i] - h1[i]));
    printf("%-20s v0-vs-v2 max|diff| = %g %s\n", sh.name, md,
           md == 0 ? "OK" : "MISMATCH");
    CHK(hipFree(dx)); CHK(hipFree(ddy)); CHK(hipFree(dw0));
    CHK(hipFree(dw1)); CHK(hipFree(ws));
    CHK(hipEventDestroy(e0)); CHK(hipEventDestroy(e1));
  }
  return 0;
}
