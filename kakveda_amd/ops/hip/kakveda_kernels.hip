// kakveda-amd CDNA4 (gfx950 / MI355X) kernels.
//
// The hot path of the failure-intelligence engine: batched cosine top-k of
// query fingerprints against the HBM3E-resident GFKB store. Replaces the
// reference's per-request TF-IDF refit + full-corpus cosine
// (reference: services/shared/similarity.py:14-20, services/gfkb/app.py:79-102).
//
// Design (MI355X-first, see /opt/skills/guides/cdna_hip_programming.md):
// - cosine_topk_partial: one workgroup owns a (128-query row tile x corpus
//   chunk). Per 128-column tile it runs an MFMA GEMM (mfma_f32_16x16x32_bf16,
//   4 waves x 64x64 output each, BK=64 K-steps, double-buffered LDS staged
//   with global_load_lds_dwordx4) and feeds the scores into an LDS-resident
//   per-row top-k list via a threshold-filtered, wave-serialised insert.
//   The per-chunk top-k lists are written out as partials.
// - topk_merge: per query row, merges the per-chunk partial lists.
// - l2normalize rows, embedding_bag: bandwidth-bound helpers (vectorised
//   bf16x8 loads per guide G13).
//
// Wavefront = 64 everywhere; LDS staging uses the lane-linear glds image
// with the XOR slot swizzle applied on the *source* address and on the
// *read* address (guide rule 21).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include <cstdint>

#define DEVINL __device__ __forceinline__

typedef __bf16 bf16_t;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

namespace {

constexpr int BM = 128;        // query rows per block
constexpr int BN = 128;        // corpus cols per tile
constexpr int BK = 64;         // K depth per LDS stage
constexpr int THREADS = 256;   // 4 waves
constexpr int KMAX = 8;        // top-k list capacity per row
constexpr int TILE_BYTES = BM * BK * 2;  // 16 KiB (BM==BN)
constexpr float NEG_INF = -1e30f;

DEVINL void glds16(const void* gsrc, void* lds_dst) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)gsrc,
      (__attribute__((address_space(3))) void*)lds_dst, 16, 0, 0);
}

// Stage one [128 rows x 64 k] bf16 tile into LDS via global_load_lds.
// LDS image is lane-linear; the 16-byte slot index is XOR-swizzled with
// (row & 7) on the global source so the fragment reads (which apply the
// same XOR) are bank-conflict-reduced (guide T2 / rule 21).
DEVINL void stage_tile(const bf16_t* __restrict__ src, int row0, int row_max,
                       long row_bytes, int ktile_byte, char* lds_tile,
                       int wid, int lane) {
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int lds_off = wid * 4096 + i * 1024;
    const int P = lds_off + lane * 16;     // this lane's physical LDS byte
    const int r = P >> 7;                  // tile row (128 B per row)
    const int s_phys = (P >> 4) & 7;       // 16-B slot within the row
    const int s_log = s_phys ^ (r & 7);    // inverse swizzle on the source
    const int gr = min(row0 + r, row_max);
    const char* gaddr =
        (const char*)src + (size_t)gr * row_bytes + ktile_byte + s_log * 16;
    glds16(gaddr, lds_tile + lds_off);
  }
}

// Read an MFMA fragment (8 contiguous bf16 along k) from a staged tile.
// row: tile-local row; slot: logical 16-B k-slot (0..7).
DEVINL bf16x8 read_frag(const char* lds_tile, int row, int slot) {
  const int s_phys = slot ^ (row & 7);
  return *(const bf16x8*)(lds_tile + row * 128 + s_phys * 16);
}

// ---------------------------------------------------------------------------
// Fused score GEMM + per-chunk top-k.
//   grid.x = nchunks, grid.y = ceil(B/128), block = 256 threads.
//   partial_score/partial_idx: [B][nchunks][KMAX]
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(THREADS, 2) void cosine_topk_partial(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ C,
    float* __restrict__ partial_score, int* __restrict__ partial_idx,
    int B, int N, int D, int chunk_tiles, int nchunks) {
  __shared__ char smem[2 * TILE_BYTES * 2            // A,B double-buffered
                       + BM * KMAX * 8               // top-k lists
                       + BM * 8];                    // row min + minpos
  char* const smem0 = smem;  // avoid static-init addrspacecast of arrays
  auto abuf = [&](int i) -> char* { return smem0 + i * TILE_BYTES; };
  auto bbuf = [&](int i) -> char* { return smem0 + (2 + i) * TILE_BYTES; };
  float* tk_score = (float*)(smem + 4 * TILE_BYTES);
  int* tk_idx = (int*)(smem + 4 * TILE_BYTES + BM * KMAX * 4);
  float* tk_min = (float*)(smem + 4 * TILE_BYTES + BM * KMAX * 8);
  int* tk_minpos = (int*)(smem + 4 * TILE_BYTES + BM * KMAX * 8 + BM * 4);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int g = lane >> 4;        // 16-lane group within the wave
  const int cl = lane & 15;
  const int wr = wid >> 1;        // wave's row quadrant (0/1)
  const int wc = wid & 1;         // wave's col quadrant (0/1)

  // XCD-aware remap (guide T1): the dispatcher places dispatch-index b on
  // XCD b%8. Give each XCD a contiguous span of corpus chunks and iterate
  // row tiles innermost, so the ~64 co-resident blocks of one XCD work on
  // the same 1-2 chunks and the corpus tile stream stays in that XCD's L2
  // instead of being re-fetched from HBM once per query row tile.
  // Requires gridDim.x % 8 == 0 (the host pads nchunks; padded chunks have
  // tiles_here <= 0 and just write -inf partials).
  const int nrt = gridDim.y;
  int chunk_id, row_tile;
  if ((gridDim.x & 7) == 0 && gridDim.x * nrt >= 512) {
    const int bid = blockIdx.x + gridDim.x * blockIdx.y;
    const int xcd = bid & 7;
    const int slot = bid >> 3;
    const int cpx = gridDim.x >> 3;  // chunks per XCD
    chunk_id = xcd * cpx + slot / nrt;
    row_tile = slot % nrt;
  } else {
    chunk_id = blockIdx.x;
    row_tile = blockIdx.y;
  }

  const int row0 = row_tile * BM;
  const long qrow_bytes = (long)D * 2;
  const int ntiles_total = (N + BN - 1) / BN;
  const int tile0 = chunk_id * chunk_tiles;
  const int tiles_here = min(chunk_tiles, ntiles_total - tile0);
  const int nkt = D / BK;

  // init top-k lists
  for (int i = tid; i < BM * KMAX; i += THREADS) {
    tk_score[i] = NEG_INF;
    tk_idx[i] = -1;
  }
  for (int i = tid; i < BM; i += THREADS) {
    tk_min[i] = NEG_INF;
    tk_minpos[i] = 0;
  }
  __syncthreads();

  for (int j = 0; j < tiles_here; ++j) {
    const int col0 = (tile0 + j) * BN;

    f32x4 acc[4][4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n) acc[m][n] = f32x4{0.f, 0.f, 0.f, 0.f};

    // prologue: stage K-tile 0
    stage_tile(Q, row0, B - 1, qrow_bytes, 0, abuf(0), wid, lane);
    stage_tile(C, col0, N - 1, qrow_bytes, 0, bbuf(0), wid, lane);
    __syncthreads();

    int cur = 0;
    for (int kt = 0; kt < nkt; ++kt) {
      if (kt + 1 < nkt) {
        const int kb = (kt + 1) * BK * 2;
        stage_tile(Q, row0, B - 1, qrow_bytes, kb, abuf(cur ^ 1), wid, lane);
        stage_tile(C, col0, N - 1, qrow_bytes, kb, bbuf(cur ^ 1), wid, lane);
      }
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        bf16x8 afrag[4], bfrag[4];
        const int slot = kk * 4 + g;
#pragma unroll
        for (int m = 0; m < 4; ++m)
          afrag[m] = read_frag(abuf(cur), wr * 64 + m * 16 + cl, slot);
#pragma unroll
        for (int n = 0; n < 4; ++n)
          bfrag[n] = read_frag(bbuf(cur), wc * 64 + n * 16 + cl, slot);
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n)
            acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[m], bfrag[n], acc[m][n], 0, 0, 0);
      }
      __syncthreads();  // drains prefetch glds; guards buffer reuse
      cur ^= 1;
    }

    // ---- top-k epilogue -------------------------------------------------
    // Stash the tile's scores into the (now idle) staging LDS, then let
    // each wave own 32 rows for the insertion pass. This keeps acc out of
    // the insertion code, which otherwise explodes register pressure
    // (measured: 147 -> 256 VGPR + 1 KB/lane scratch when the unrolled
    // insert reads acc directly).
#ifdef KAKVEDA_NO_EPILOGUE
    // ablation build: keep acc live, skip the top-k insert (guide rule 17)
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n)
        asm volatile("" ::"v"(acc[m][n]));
    if (false)
#endif
    {
      float* stile = (float*)smem0;  // [128][128] f32 over the staging bufs
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
#pragma unroll
          for (int reg = 0; reg < 4; ++reg) {
            const int row = wr * 64 + m * 16 + g * 4 + reg;
            const int col = wc * 64 + n * 16 + cl;
            stile[row * 128 + col] = acc[m][n][reg];
          }
      __syncthreads();

      volatile float* vmin = tk_min;
      volatile float* vscore = tk_score;
      volatile int* vidx = tk_idx;
      volatile int* vpos = tk_minpos;

      for (int rr = 0; rr < 32; ++rr) {
        const int row = wid * 32 + rr;
        const float rmin0 = vmin[row];
        const float v0 = stile[row * 128 + lane];
        const float v1 = stile[row * 128 + 64 + lane];
        const int gc0 = col0 + lane;
        const int gc1 = col0 + 64 + lane;
        const bool a0 = (gc0 < N) & (v0 > rmin0);
        const bool a1 = (gc1 < N) & (v1 > rmin0);
        unsigned long long ball = __ballot(a0 | a1);
        while (ball) {
          const int leader = __ffsll((unsigned long long)ball) - 1;
          if (lane == leader) {
#pragma unroll
            for (int h = 0; h < 2; ++h) {
              const float v = h ? v1 : v0;
              const int gc = h ? gc1 : gc0;
              const bool a = h ? a1 : a0;
              if (a && v > vmin[row]) {
                const int p = vpos[row];
                vscore[row * KMAX + p] = v;
                vidx[row * KMAX + p] = gc;
                float mn = vscore[row * KMAX];
                int mp = 0;
#pragma unroll
                for (int q = 1; q < KMAX; ++q) {
                  const float s = vscore[row * KMAX + q];
                  if (s < mn) { mn = s; mp = q; }
                }
                vmin[row] = mn;
                vpos[row] = mp;
              }
            }
          }
          ball &= ball - 1;
        }
      }
      __syncthreads();  // lists settled before next tile reuses the LDS
    }
  }

  // write partials: [B][nchunks][KMAX]
  if (tid < BM) {
    const int grow = row0 + tid;
    if (grow < B) {
      const size_t base = ((size_t)grow * nchunks + chunk_id) * KMAX;
#pragma unroll
      for (int q = 0; q < KMAX; ++q) {
        partial_score[base + q] = tk_score[tid * KMAX + q];
        partial_idx[base + q] = tk_idx[tid * KMAX + q];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Merge per-chunk partial lists -> final sorted top-k per query row.
//   grid = B blocks, 256 threads.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(THREADS) void topk_merge(
    const float* __restrict__ partial_score, const int* __restrict__ partial_idx,
    float* __restrict__ out_score, long* __restrict__ out_idx,
    int nchunks, int k) {
  __shared__ float sc[THREADS * KMAX];
  __shared__ int si[THREADS * KMAX];

  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const int total = nchunks * KMAX;
  const size_t base = (size_t)row * total;

  float ls[KMAX];
  int li[KMAX];
#pragma unroll
  for (int q = 0; q < KMAX; ++q) { ls[q] = NEG_INF; li[q] = -1; }
  float lmin = NEG_INF;

  for (int e = tid; e < total; e += THREADS) {
    const float s = partial_score[base + e];
    if (s > lmin) {
      // replace current min
      int mp = 0;
      float mn = ls[0];
#pragma unroll
      for (int q = 1; q < KMAX; ++q)
        if (ls[q] < mn) { mn = ls[q]; mp = q; }
#pragma unroll
      for (int q = 0; q < KMAX; ++q)
        if (q == mp) { ls[q] = s; li[q] = partial_idx[base + e]; }
      lmin = ls[0];
#pragma unroll
      for (int q = 1; q < KMAX; ++q) lmin = fminf(lmin, ls[q]);
    }
  }
#pragma unroll
  for (int q = 0; q < KMAX; ++q) {
    sc[tid * KMAX + q] = ls[q];
    si[tid * KMAX + q] = li[q];
  }
  __syncthreads();

  if (tid == 0) {
    float fs[KMAX];
    int fi[KMAX];
#pragma unroll
    for (int q = 0; q < KMAX; ++q) { fs[q] = NEG_INF; fi[q] = -1; }
    for (int e = 0; e < THREADS * KMAX; ++e) {
      const float s = sc[e];
      int mp = 0;
      float mn = fs[0];
#pragma unroll
      for (int q = 1; q < KMAX; ++q)
        if (fs[q] < mn) { mn = fs[q]; mp = q; }
      if (s > mn) {
#pragma unroll
        for (int q = 0; q < KMAX; ++q)
          if (q == mp) { fs[q] = s; fi[q] = si[e]; }
      }
    }
    // sort descending (insertion sort, KMAX small)
#pragma unroll
    for (int a = 1; a < KMAX; ++a) {
      const float s = fs[a];
      const int ix = fi[a];
      int b = a - 1;
      for (; b >= 0 && fs[b] < s; --b) { fs[b + 1] = fs[b]; fi[b + 1] = fi[b]; }
      fs[b + 1] = s;
      fi[b + 1] = ix;
    }
    for (int q = 0; q < k; ++q) {
      out_score[(size_t)row * k + q] = fs[q] <= NEG_INF ? -INFINITY : fs[q];
      out_idx[(size_t)row * k + q] = fi[q];
    }
  }
}

// ---------------------------------------------------------------------------
// Row-wise L2 normalisation, in place. One wave per row, bf16x8 loads.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(THREADS) void l2normalize_rows(
    bf16_t* __restrict__ data, int start_row, int nrows, int D) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int waves_per_grid = gridDim.x * (THREADS / 64);

  for (int r = blockIdx.x * (THREADS / 64) + wave; r < nrows;
       r += waves_per_grid) {
    bf16_t* row = data + (size_t)(start_row + r) * D;
    float ss = 0.f;
    for (int d = lane * 8; d < D; d += 64 * 8) {
      const bf16x8 v = *(const bf16x8*)(row + d);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const float f = (float)v[i];
        ss += f * f;
      }
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) ss += __shfl_xor(ss, off, 64);
    const float scale = (ss > 1e-24f) ? rsqrtf(ss) : 0.f;
    for (int d = lane * 8; d < D; d += 64 * 8) {
      bf16x8 v = *(const bf16x8*)(row + d);
#pragma unroll
      for (int i = 0; i < 8; ++i) v[i] = (bf16_t)((float)v[i] * scale);
      *(bf16x8*)(row + d) = v;
    }
  }
}

// ---------------------------------------------------------------------------
// Weighted embedding bag: out[b] = sum_l w[b,l] * table[idx[b,l]].
//   table [V][D] bf16, idx [B][L] i32, w [B][L] f32 -> out [B][D] f32.
//   One block per bag; threads stride over D.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(THREADS) void embedding_bag_kernel(
    const bf16_t* __restrict__ table, const int* __restrict__ idx,
    const float* __restrict__ w, float* __restrict__ out, int L, int D,
    int V) {
  __shared__ int s_idx[128];
  __shared__ float s_w[128];
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  for (int l = tid; l < L; l += THREADS) {
    s_idx[l] = idx[(size_t)b * L + l];
    s_w[l] = w[(size_t)b * L + l];
  }
  __syncthreads();

  for (int d = tid; d < D; d += THREADS) {
    float acc = 0.f;
    for (int l = 0; l < L; ++l) {
      const float wl = s_w[l];
      if (wl != 0.f) {
        const int ix = s_idx[l];
        if (ix >= 0 && ix < V) acc += wl * (float)table[(size_t)ix * D + d];
      }
    }
    out[(size_t)b * D + d] = acc;
  }
}

}  // namespace

// ===========================================================================
// Torch bindings
// ===========================================================================

static void check_bf16_2d(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.dim() == 2 && t.is_contiguous(), name, " must be contiguous 2-D");
}

std::tuple<torch::Tensor, torch::Tensor> cosine_topk(
    torch::Tensor queries, torch::Tensor corpus, int64_t k, int64_t valid_n) {
  check_bf16_2d(queries, "queries");
  check_bf16_2d(corpus, "corpus");
  const int B = queries.size(0);
  const int D = queries.size(1);
  const int N = (int)valid_n;
  TORCH_CHECK(corpus.size(1) == D, "dim mismatch");
  TORCH_CHECK(N >= 1 && N <= corpus.size(0), "valid_n out of range");
  TORCH_CHECK(D % BK == 0, "D must be a multiple of ", BK);
  TORCH_CHECK(k >= 1 && k <= KMAX, "k must be in [1,", KMAX, "]");

  const int row_tiles = (B + BM - 1) / BM;
  const int ntiles = (N + BN - 1) / BN;
  // size chunks so the grid comfortably oversubscribes 256 CUs
  long want = ((long)ntiles * row_tiles + 2047) / 2048;
  const int chunk_tiles = (int)std::max(4L, std::min(want, 128L));
  // pad the chunk count to a multiple of 8 so the in-kernel XCD remap is
  // bijective; padded chunks have no tiles and emit -inf partials.
  const int nchunks = ((ntiles + chunk_tiles - 1) / chunk_tiles + 7) & ~7;

  auto opts_f = torch::TensorOptions().dtype(torch::kFloat32).device(queries.device());
  auto opts_i = torch::TensorOptions().dtype(torch::kInt32).device(queries.device());
  auto pscore = torch::empty({(long)B * nchunks * KMAX}, opts_f);
  auto pidx = torch::empty({(long)B * nchunks * KMAX}, opts_i);
  auto out_score = torch::empty({B, k}, opts_f);
  auto out_idx = torch::empty({B, k}, torch::TensorOptions().dtype(torch::kInt64).device(queries.device()));

  auto stream = c10::hip::getCurrentHIPStream();
  dim3 grid(nchunks, row_tiles);
  hipLaunchKernelGGL(cosine_topk_partial, grid, dim3(THREADS), 0, stream.stream(),
                     (const bf16_t*)queries.data_ptr(), (const bf16_t*)corpus.data_ptr(),
                     pscore.data_ptr<float>(), pidx.data_ptr<int>(),
                     B, N, D, chunk_tiles, nchunks);
  hipLaunchKernelGGL(topk_merge, dim3(B), dim3(THREADS), 0, stream.stream(),
                     pscore.data_ptr<float>(), pidx.data_ptr<int>(),
                     out_score.data_ptr<float>(), (long*)out_idx.data_ptr<int64_t>(),
                     nchunks, (int)k);
  return {out_score, out_idx};
}

void l2normalize_(torch::Tensor t, int64_t start_row, int64_t end_row) {
  check_bf16_2d(t, "t");
  const int nrows = (int)(end_row - start_row);
  if (nrows <= 0) return;
  const int D = t.size(1);
  TORCH_CHECK(D % 8 == 0, "D must be a multiple of 8");
  auto stream = c10::hip::getCurrentHIPStream();
  const int blocks = std::min((nrows + 3) / 4, 2048);
  hipLaunchKernelGGL(l2normalize_rows, dim3(blocks), dim3(THREADS), 0,
                     stream.stream(), (bf16_t*)t.data_ptr(), (int)start_row,
                     nrows, D);
}

torch::Tensor embedding_bag(torch::Tensor table, torch::Tensor idx, torch::Tensor w) {
  check_bf16_2d(table, "table");
  TORCH_CHECK(idx.is_cuda() && idx.scalar_type() == torch::kInt32 && idx.dim() == 2 && idx.is_contiguous(), "idx must be contiguous 2-D i32 on GPU");
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == torch::kFloat32 && w.is_contiguous(), "w must be contiguous f32 on GPU");
  const int B = idx.size(0);
  const int L = idx.size(1);
  const int D = table.size(1);
  const int V = table.size(0);
  TORCH_CHECK(L <= 128, "L must be <= 128");
  TORCH_CHECK(w.size(0) == B && w.size(1) == L, "w shape mismatch");
  auto out = torch::empty({B, D}, torch::TensorOptions().dtype(torch::kFloat32).device(table.device()));
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(embedding_bag_kernel, dim3(B), dim3(THREADS), 0,
                     stream.stream(), (const bf16_t*)table.data_ptr(),
                     idx.data_ptr<int>(), w.data_ptr<float>(),
                     out.data_ptr<float>(), L, D, V);
  return out;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("cosine_topk", &cosine_topk, "fused cosine top-k (MFMA + LDS top-k)");
  m.def("l2normalize_", &l2normalize_, "in-place row L2 normalisation");
  m.def("embedding_bag", &embedding_bag, "weighted embedding bag");
  m.attr("KMAX") = KMAX;
}
