// Standalone perf probe for the cosine-topk kernel (no torch, no python).
// Build:  hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/knn_probe.hip -o gpurun_out/knn_probe
//         (add -DKAKVEDA_NO_EPILOGUE for the GEMM-only ablation)
// Run:    ./knn_probe [B] [N] [iters]
// Prints ms/iter and effective TFLOP/s of the score GEMM.

#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdlib>
#include <vector>
#include <algorithm>

#include "../kakveda_amd/ops/hip/kernels_impl.h"

using namespace kakveda;

#define HIP_CHECK(x)                                                    \
  do {                                                                  \
    hipError_t e = (x);                                                 \
    if (e != hipSuccess) {                                              \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e),  \
              __FILE__, __LINE__);                                      \
      exit(1);                                                          \
    }                                                                   \
  } while (0)

__global__ void fill_rand(bf16_t* p, size_t n, unsigned seed) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  const size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    unsigned h = (unsigned)(i * 2654435761u) ^ seed;
    h ^= h >> 13; h *= 0x5bd1e995u; h ^= h >> 15;
    const float f = ((float)(h & 0xFFFF) / 65536.0f - 0.5f) * 0.07217f; // ~unit rows
    p[i] = (bf16_t)f;
  }
}

int main(int argc, char** argv) {
  const int B = argc > 1 ? atoi(argv[1]) : 4096;
  const long N = argc > 2 ? atol(argv[2]) : 2000000;
  const int iters = argc > 3 ? atoi(argv[3]) : 10;
  const int D = 768, k = 5;

  bf16_t *Q, *C;
  HIP_CHECK(hipMalloc(&Q, (size_t)B * D * 2));
  HIP_CHECK(hipMalloc(&C, (size_t)N * D * 2));
  hipLaunchKernelGGL(fill_rand, dim3(2048), dim3(256), 0, 0, Q, (size_t)B * D, 1u);
  hipLaunchKernelGGL(fill_rand, dim3(2048), dim3(256), 0, 0, C, (size_t)N * D, 2u);

  const int row_tiles = (B + BM - 1) / BM;
  const int ntiles = (int)((N + BN - 1) / BN);
  long want = ((long)ntiles * row_tiles + 2047) / 2048;
  const int chunk_tiles = (int)std::max(4L, std::min(want, 128L));
  const int nchunks = ((ntiles + chunk_tiles - 1) / chunk_tiles + 7) & ~7;

  float* pscore;
  int* pidx;
  float* oscore;
  long* oidx;
  HIP_CHECK(hipMalloc(&pscore, (size_t)B * nchunks * KMAX * 4));
  HIP_CHECK(hipMalloc(&pidx, (size_t)B * nchunks * KMAX * 4));
  HIP_CHECK(hipMalloc(&oscore, (size_t)B * k * 4));
  HIP_CHECK(hipMalloc(&oidx, (size_t)B * k * 8));
  unsigned* rowthr;
  HIP_CHECK(hipMalloc(&rowthr, (size_t)B * 4));

  dim3 grid(nchunks, row_tiles);
  // 256-tile variant geometry
  const int row_tiles2 = (B + BM8 - 1) / BM8;
  const int ntiles2 = (int)((N + BN8 - 1) / BN8);
  long want2 = ((long)ntiles2 * row_tiles2 + 511) / 512;
  const int chunk_tiles2 = (int)std::max(4L, std::min(want2, 128L));
  const int nchunks2 = ((ntiles2 + chunk_tiles2 - 1) / chunk_tiles2 + 7) & ~7;
  float* pscore2;
  int* pidx2;
  float* slab2;
  HIP_CHECK(hipMalloc(&pscore2, (size_t)B * nchunks2 * KMAX * 4));
  HIP_CHECK(hipMalloc(&pidx2, (size_t)B * nchunks2 * KMAX * 4));
  HIP_CHECK(hipMalloc(&slab2, (size_t)nchunks2 * row_tiles2 * BM8 * BN8 * 4));
  dim3 grid2(nchunks2, row_tiles2);
  auto run_mode = [&](int mode) {
    // modes 12/13: "warm" variants — keep rowthr from the previous
    // iteration (same data, so thresholds converge to the exact per-row
    // k-th best): measures the ideal-threshold-warming ceiling.
    if (mode != 12 && mode != 13 && mode != 15 && mode != 19)
      hipLaunchKernelGGL(init_rowthr, dim3((B + 255) / 256), dim3(256), 0, 0,
                         rowthr, B);
    if (mode == 0)
      hipLaunchKernelGGL((cosine_topk_partial_t<0>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 1)
      hipLaunchKernelGGL((cosine_topk_partial_t<1>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 2)
      hipLaunchKernelGGL((cosine_topk_partial_t<2>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 7)
      hipLaunchKernelGGL((cosine_topk_partial_t<0, 1>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 8)
      hipLaunchKernelGGL((cosine_topk_partial_t<1, 1>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 10)
      hipLaunchKernelGGL((cosine_topk_partial_t<7>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 11)
      hipLaunchKernelGGL((cosine_topk_partial_t<8>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 12)
      hipLaunchKernelGGL((cosine_topk_partial_t<0>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 13)
      hipLaunchKernelGGL((cosine_topk_partial_t<8>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 14)
      hipLaunchKernelGGL((cosine_topk_partial_t<9>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 15) // warm variant of 9
      hipLaunchKernelGGL((cosine_topk_partial_t<9>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 16)
      hipLaunchKernelGGL((cosine_topk_partial_t<10>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 17)
      hipLaunchKernelGGL((cosine_topk_partial_t<4>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 18 || mode == 19)
      hipLaunchKernelGGL((cosine_topk_partial_t<11>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 20)
      hipLaunchKernelGGL((cosine_topk_partial8p_t<6>), grid2, dim3(THREADS8), 0, 0,
                         Q, C, pscore2, pidx2, B, (int)N, D, chunk_tiles2, nchunks2, rowthr, (unsigned long long*)nullptr);
    else if (mode == 21)
      hipLaunchKernelGGL((cosine_topk_partial8p_t<7>), grid2, dim3(THREADS8), 0, 0,
                         Q, C, pscore2, pidx2, B, (int)N, D, chunk_tiles2, nchunks2, rowthr, (unsigned long long*)nullptr);
    else if (mode == 22)
      hipLaunchKernelGGL((cosine_topk_partial_t<12>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 6)
      hipLaunchKernelGGL((cosine_topk_partial_t<6>), grid, dim3(THREADS), 0, 0,
                         Q, C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks, rowthr, (unsigned long long*)nullptr);
    else if (mode == 9)
      hipLaunchKernelGGL((cosine_topk_partial8p_t<2>), grid2, dim3(THREADS8), 0, 0,
                         Q, C, pscore2, pidx2, B, (int)N, D, chunk_tiles2, nchunks2, rowthr, (unsigned long long*)nullptr);
    else if (mode == 4)
      hipLaunchKernelGGL((cosine_topk_partial8p_t<5>), grid2, dim3(THREADS8), 0, 0,
                         Q, C, pscore2, pidx2, B, (int)N, D, chunk_tiles2, nchunks2, rowthr, (unsigned long long*)nullptr, slab2);
    else if (mode == 3)
      hipLaunchKernelGGL((cosine_topk_partial8p_t<0>), grid2, dim3(THREADS8), 0, 0,
                         Q, C, pscore2, pidx2, B, (int)N, D, chunk_tiles2, nchunks2, rowthr, (unsigned long long*)nullptr);
    else
      hipLaunchKernelGGL((cosine_topk_partial8p_t<1>), grid2, dim3(THREADS8), 0, 0,
                         Q, C, pscore2, pidx2, B, (int)N, D, chunk_tiles2, nchunks2, rowthr, (unsigned long long*)nullptr);
    hipError_t le = hipGetLastError();
    if (le != hipSuccess)
      fprintf(stderr, "launch error (mode %d): %s\n", mode, hipGetErrorString(le));
  };

  const int NM = 6;
  const int warm_modes[NM] = {18, 22, 1, 0, 14, 20};
  for (int mi = 0; mi < NM; ++mi) run_mode(warm_modes[mi]);
  HIP_CHECK(hipDeviceSynchronize());

  const char* names[23] = {"full128", "gemm128", "precheck128", "-", "slab8p", "gemm8p", "full128-bl", "full128-bk32", "gemm128-bk32", "precheck8p", "dfr128", "rege128", "fullwarm128", "regewarm128", "fast128", "fastwarm128", "faststorm128", "argmax128", "fastbl128", "fastblwarm128", "fastbl8p", "queue8p", "fastblprio128"};
  const int modes[NM] = {18, 22, 1, 0, 14, 20};
  std::vector<std::vector<float>> ms(23);
  hipEvent_t t0, t1;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  for (int it = 0; it < iters; ++it) {
    for (int mi = 0; mi < NM; ++mi) {
      const int m = modes[mi];
      HIP_CHECK(hipEventRecord(t0));
      run_mode(m);
      HIP_CHECK(hipEventRecord(t1));
      HIP_CHECK(hipEventSynchronize(t1));
      float x;
      HIP_CHECK(hipEventElapsedTime(&x, t0, t1));
      ms[m].push_back(x);
    }
  }
  // stats pass
  unsigned long long* dstats;
  HIP_CHECK(hipMalloc(&dstats, 4 * 8));
  HIP_CHECK(hipMemset(dstats, 0, 4 * 8));
  hipLaunchKernelGGL(init_rowthr, dim3((B + 255) / 256), dim3(256), 0, 0,
                     rowthr, B);
  hipLaunchKernelGGL((cosine_topk_partial_t<3>), grid, dim3(THREADS), 0, 0, Q,
                     C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks,
                     rowthr, dstats);
  HIP_CHECK(hipDeviceSynchronize());
  unsigned long long hstats[4];
  HIP_CHECK(hipMemcpy(hstats, dstats, 32, hipMemcpyDeviceToHost));
  const double tiles = (double)row_tiles * ntiles;
  printf("stats: stamped=%llu (%.2f/tile) inserts=%llu (%.1f/row/chunk)\n",
         hstats[0], hstats[0] / tiles, hstats[1],
         hstats[1] / ((double)B * nchunks));
  // warm stats: rowthr already converged from the pass above -> counts
  // the steady-state (bootstrap-free) qualifying rate.
  HIP_CHECK(hipMemset(dstats, 0, 4 * 8));
  hipLaunchKernelGGL((cosine_topk_partial_t<3>), grid, dim3(THREADS), 0, 0, Q,
                     C, pscore, pidx, B, (int)N, D, chunk_tiles, nchunks,
                     rowthr, dstats);
  HIP_CHECK(hipDeviceSynchronize());
  HIP_CHECK(hipMemcpy(hstats, dstats, 32, hipMemcpyDeviceToHost));
  printf("warm stats: stamped=%llu (%.2f/tile)\n", hstats[0],
         hstats[0] / tiles);
  for (int mi = 0; mi < NM; ++mi) {
    const int m = modes[mi];
    std::sort(ms[m].begin(), ms[m].end());
    const float med = ms[m][ms[m].size() / 2];
    const float mn = ms[m][0];
    const double tf = 2.0 * B * (double)N * D / (med * 1e-3) / 1e12;
    printf("%-15s B=%d N=%ld med=%.3f ms min=%.3f  %.1f TF\n", names[m], B, N,
           med, mn, tf);
  }
  return 0;
}
