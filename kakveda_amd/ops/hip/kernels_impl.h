// Device-code implementation shared by the torch extension and the
// standalone perf probe (tools/knn_probe.hip). Pure HIP, no torch.
#pragma once
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
#include <cstdint>

#define DEVINL __device__ __forceinline__

typedef __bf16 bf16_t;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

namespace kakveda {

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

// Order-preserving f32 <-> u32 encoding so float thresholds can be shared
// through u32 atomicMax (x < y  <=>  enc(x) < enc(y)).
DEVINL unsigned enc_f32(float x) {
  const unsigned b = __float_as_uint(x);
  return (b & 0x80000000u) ? ~b : (b | 0x80000000u);
}
DEVINL float dec_f32(unsigned u) {
  return __uint_as_float((u & 0x80000000u) ? (u ^ 0x80000000u) : ~u);
}

__global__ void init_rowthr(unsigned* thr, int n) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) thr[i] = enc_f32(NEG_INF);
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

// Generalised variants: tiles of [128 rows][NKK*32 k] (row = NKK*64 bytes).
template <int NKK>
DEVINL void stage_tile_n(const bf16_t* __restrict__ src, int row0, int row_max,
                         long row_bytes, int ktile_byte, char* lds_tile,
                         int wid, int lane) {
  constexpr int ROWB = NKK * 64;     // bytes per tile row
  constexpr int SLOTS = NKK * 4;     // 16-B slots per row
#pragma unroll
  for (int i = 0; i < 2 * NKK; ++i) {
    const int lds_off = wid * (2 * NKK) * 1024 + i * 1024;
    const int P = lds_off + lane * 16;
    const int r = P / ROWB;
    const int s_phys = (P >> 4) % SLOTS;
    const int s_log = NKK == 2 ? (s_phys ^ (r & 7)) : (s_phys ^ ((r >> 2) & 3));
    const int gr = min(row0 + r, row_max);
    const char* gaddr =
        (const char*)src + (size_t)gr * row_bytes + ktile_byte + s_log * 16;
    glds16(gaddr, lds_tile + lds_off);
  }
}

template <int NKK>
DEVINL bf16x8 read_frag_n(const char* lds_tile, int row, int slot) {
  const int s_phys = NKK == 2 ? (slot ^ (row & 7)) : (slot ^ ((row >> 2) & 3));
  return *(const bf16x8*)(lds_tile + row * (NKK * 64) + s_phys * 16);
}


// Extraction for one qualifying (row, col-half): deliberately __noinline__
// so the 16 unrolled epilogue bodies don't get their candidate values
// hoisted live simultaneously (measured: inlining costs 100+ VGPRs and
// ~850 B/lane of scratch spill).
// Ballot-leader variant: no per-insert wave argmax reduce — each loop
// iteration picks the lowest qualifying lane, which inserts its own best
// candidate. Fewer cross-lane ops per insert; insertion order is
// arbitrary (list semantics identical).
template <bool PUBLISH>
__device__ __noinline__ void topk_extract_group_bl(
    volatile float* vsc, volatile int* vix, int lbase, float rwarm,
    float w0, float w1, float w2, float w3, int colb, int N_unused, int lane,
    int g, unsigned* rowthr, int growp1) {
  float rmin = fmaxf(vsc[lbase], rwarm);
  while (true) {
    float b = w0;
    int bn = 0;
    if (w1 > b) { b = w1; bn = 1; }
    if (w2 > b) { b = w2; bn = 2; }
    if (w3 > b) { b = w3; bn = 3; }
    const unsigned long long ball =
        __ballot(b > rmin) & (0xFFFFull << (g * 16));
    if (!ball) break;
    const int leader = __ffsll(ball) - 1;
    if (lane == leader) {
      float nmn = b;
      int nmp = 0;
#pragma unroll
      for (int q = 1; q < KMAX; ++q) {
        const float s = vsc[lbase + q];
        if (s < nmn) { nmn = s; nmp = q; }
      }
      const int gcol = colb + bn * 16;
      if (nmp == 0) {
        vsc[lbase] = b;
        vix[lbase] = gcol;
      } else {
        const int mi2 = vix[lbase + nmp];
        vsc[lbase] = nmn;
        vix[lbase] = mi2;
        vsc[lbase + nmp] = b;
        vix[lbase + nmp] = gcol;
      }
      if (bn == 0) w0 = NEG_INF;
      else if (bn == 1) w1 = NEG_INF;
      else if (bn == 2) w2 = NEG_INF;
      else w3 = NEG_INF;
      if (PUBLISH && rowthr != nullptr && nmn > NEG_INF && growp1 > 0)
        atomicMax(&rowthr[growp1 - 1], enc_f32(nmn));
    }
    rmin = fmaxf(vsc[lbase], rwarm);  // same-wave LDS program order
  }
}

template <bool PUBLISH>
__device__ __noinline__ void topk_extract_group(
    volatile float* vsc, volatile int* vix, int lbase, float rwarm,
    float w0, float w1, float w2, float w3, int colb, int N_unused, int lane,
    int g, unsigned* rowthr, int growp1) {
  float rmin = fmaxf(vsc[lbase], rwarm);
  while (true) {
    float b = w0;
    int bn = 0;
    if (w1 > b) { b = w1; bn = 1; }
    if (w2 > b) { b = w2; bn = 2; }
    if (w3 > b) { b = w3; bn = 3; }
    float mv = (b > rmin) ? b : NEG_INF;
#pragma unroll
    for (int off = 1; off < 16; off <<= 1)
      mv = fmaxf(mv, __shfl_xor(mv, off, 64));
    if (mv <= NEG_INF) break;
    const unsigned long long winners =
        __ballot(b == mv && b > rmin) & (0xFFFFull << (g * 16));
    const int leader = __ffsll(winners) - 1;
    if (lane == leader) {
      float nmn = mv;
      int nmp = 0;
#pragma unroll
      for (int q = 1; q < KMAX; ++q) {
        const float s = vsc[lbase + q];
        if (s < nmn) { nmn = s; nmp = q; }
      }
      const int gcol = colb + bn * 16;
      if (nmp == 0) {
        vsc[lbase] = mv;
        vix[lbase] = gcol;
      } else {
        const int mi2 = vix[lbase + nmp];
        vsc[lbase] = nmn;
        vix[lbase] = mi2;
        vsc[lbase + nmp] = mv;
        vix[lbase + nmp] = gcol;
      }
      if (bn == 0) w0 = NEG_INF;
      else if (bn == 1) w1 = NEG_INF;
      else if (bn == 2) w2 = NEG_INF;
      else w3 = NEG_INF;
      if (PUBLISH && rowthr != nullptr && nmn > NEG_INF && growp1 > 0)
        atomicMax(&rowthr[growp1 - 1], enc_f32(nmn));
    }
    rmin = fmaxf(vsc[lbase], rwarm);  // same-wave LDS order
  }
}

// ---------------------------------------------------------------------------
// Fused score GEMM + per-chunk top-k.
//   grid.x = nchunks, grid.y = ceil(B/128), block = 256 threads.
//   partial_score/partial_idx: [B][nchunks][KMAX]
// ---------------------------------------------------------------------------
// EPI_MODE: 0 full top-k (volatile-LDS thresholds, immediate extraction),
// 1 GEMM-only ablation, 2 pre-check only (NOTE: its empty taken-branch
// lets the compiler dead-code-eliminate the MFMAs — not a valid ceiling;
// use 1), 3 full+stats, 4 argmax (k=1 fast path: per-row max, no
// lists/extraction), 6 ballot-leader extraction, 7 deferred
// register-threshold epilogue (extraction after the tile barrier;
// measured slower), 8 register-cached thresholds + call-only extraction,
// 9 = 8 + inline single-insert fast path, 10 = 9 + bootstrap bypass
// (parity), 11 = 9 with a masked-__ballot pre-check instead of the
// 4-deep shfl_xor max reduce — the PRODUCTION DEFAULT (754 TF vs 642
// for mode 0 at B=4096 x N=2M; see profiles/knn_kernel_history.md).
template <int EPI_MODE, int NKK = 2>  // NKK: 32-deep K steps per LDS stage (2 -> BK=64)
__global__ __launch_bounds__(THREADS, 4 - NKK) void cosine_topk_partial_t(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ C,
    float* __restrict__ partial_score, int* __restrict__ partial_idx,
    int B, int N, int D, int chunk_tiles, int nchunks,
    unsigned* rowthr = nullptr, unsigned long long* stats = nullptr) {
  // LDS: 4 x 16 KiB staging (A,B double-buffered) + per-(row, col-half)
  // private top-k lists. 80 KiB total -> 2 blocks/CU. The lists are
  // private to the one wave that computes that (row-half, col-half), so
  // the epilogue needs NO barriers and overlaps the next tile's staging.
  constexpr int TB = BM * NKK * 32 * 2;  // one staged tile
  __shared__ char smem[2 * TB * 2 + 2 * BM * KMAX * 8];
  char* const smem0 = smem;  // avoid static-init addrspacecast of arrays
  auto abuf = [&](int i) -> char* { return smem0 + i * TB; };
  auto bbuf = [&](int i) -> char* { return smem0 + (2 + i) * TB; };
  // list layout: scores[wc][row][KMAX] then idx[wc][row][KMAX];
  // invariant: slot 0 of each list holds that list's MINIMUM.
  float* lsc = (float*)(smem + 4 * TB);
  int* lix = (int*)(smem + 4 * TB + 2 * BM * KMAX * 4);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int g = lane >> 4;        // 16-lane group within the wave
  const int cl = lane & 15;
  const int wr = wid >> 1;        // wave's row half (0/1)
  const int wc = wid & 1;         // wave's col half (0/1)

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
  const int nkt = D / (NKK * 32);

  // init lists (-inf scores; slot 0 is the min by construction)
  for (int i = tid; i < 2 * BM * KMAX; i += THREADS) {
    lsc[i] = NEG_INF;
    lix[i] = -1;
  }
  __syncthreads();

  // prologue: stage tile 0's first K-tile
  if (tiles_here > 0) {
    stage_tile(Q, row0, B - 1, qrow_bytes, 0, abuf(0), wid, lane);
    stage_tile(C, tile0 * BN, N - 1, qrow_bytes, 0, bbuf(0), wid, lane);
  }
  __syncthreads();
  int cur = 0;

  // EPI_MODE 7: per-row pruning threshold cached in a register. Lane l of
  // each wave owns max(list-min, warm) for row wr*64+l of col-half wc, so
  // the pre-check needs no LDS reads at all (the volatile list reads at
  // the 16 unrolled call boundaries otherwise serialize ~70 lgkmcnt(0)
  // waits per tile per wave — measured 650 vs 884 TF full-vs-precheck).
  float rmin_reg = NEG_INF;
  (void)rmin_reg;

  for (int j = 0; j < tiles_here; ++j) {
    const int col0 = (tile0 + j) * BN;

    // per-lane warm thresholds: lane l caches rowthr for row wr*64+l; the
    // pre-check broadcasts the right lane's value with one shfl. Published
    // thresholds from other blocks prune chunk bootstraps to ~nothing.
    float warm = NEG_INF;
    if (EPI_MODE != 1 && rowthr != nullptr && row0 + wr * 64 + lane < B)
      warm = dec_f32(rowthr[row0 + wr * 64 + lane]);
    if constexpr (EPI_MODE >= 7 && EPI_MODE <= 12)
      rmin_reg = fmaxf(rmin_reg, warm);

    f32x4 acc[4][4];
#pragma unroll
    for (int m = 0; m < 4; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n) acc[m][n] = f32x4{0.f, 0.f, 0.f, 0.f};

    for (int kt = 0; kt < nkt; ++kt) {
      // seamless cross-tile pipeline: the last K-step of tile j prefetches
      // tile j+1's first K-tile, so the epilogue below runs while that
      // staging is in flight.
      if (kt + 1 < nkt) {
        const int kb = (kt + 1) * NKK * 64;
        stage_tile_n<NKK>(Q, row0, B - 1, qrow_bytes, kb, abuf(cur ^ 1), wid, lane);
        stage_tile_n<NKK>(C, col0, N - 1, qrow_bytes, kb, bbuf(cur ^ 1), wid, lane);
      } else if (j + 1 < tiles_here) {
        stage_tile_n<NKK>(Q, row0, B - 1, qrow_bytes, 0, abuf(cur ^ 1), wid, lane);
        stage_tile_n<NKK>(C, col0 + BN, N - 1, qrow_bytes, 0, bbuf(cur ^ 1), wid, lane);
      }
#pragma unroll
      for (int kk = 0; kk < NKK; ++kk) {
        bf16x8 afrag[4], bfrag[4];
        const int slot = kk * 4 + g;
#pragma unroll
        for (int m = 0; m < 4; ++m)
          afrag[m] = read_frag_n<NKK>(abuf(cur), wr * 64 + m * 16 + cl, slot);
#pragma unroll
        for (int n = 0; n < 4; ++n)
          bfrag[n] = read_frag_n<NKK>(bbuf(cur), wc * 64 + n * 16 + cl, slot);
        // EPI_MODE 12 experiment: raise wave priority through the MFMA
        // burst (8p-style s_setprio) so the co-resident wave's VALU work
        // interleaves under it
        if constexpr (EPI_MODE == 12) __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n)
            acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[m], bfrag[n], acc[m][n], 0, 0, 0);
        if constexpr (EPI_MODE == 12) __builtin_amdgcn_s_setprio(0);
      }
      if (kt + 1 < nkt) {
        __syncthreads();  // drains prefetch glds; guards buffer reuse
        cur ^= 1;
      }
      // after the LAST K-step the barrier is deferred below the epilogue,
      // overlapping list maintenance with the next tile's staging.
    }

    // ---- top-k epilogue (barrier-free) ----------------------------------
    // Each (row, col-half) list is owned by exactly one wave. Per (m,reg)
    // row, a 4-step intra-group shfl computes the row-half max from the
    // accumulators; the (rare) qualifying rows extract their candidates in
    // descending order, one leader lane updating the private LDS list.
    unsigned qmask = 0;
    (void)qmask;
    if constexpr (EPI_MODE == 1) {
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          asm volatile("" ::"v"(acc[m][n]));
    } else if constexpr (EPI_MODE == 7) {
      // pre-check only here (registers + cross-lane ops, no LDS, no
      // calls); qualifying (m,reg) groups set a bit and extract after
      // the tile barrier below, where vmcnt is naturally drained so the
      // noinline callee's conservative entry wait costs nothing.
      const int colb = col0 + wc * 64 + cl;
#pragma unroll
      for (int m = 0; m < 4; ++m) {
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const float rwall = __shfl(rmin_reg, m * 16 + g * 4 + reg, 64);
          float w0 = (colb + 0 < N) ? acc[m][0][reg] : NEG_INF;
          float w1 = (colb + 16 < N) ? acc[m][1][reg] : NEG_INF;
          float w2 = (colb + 32 < N) ? acc[m][2][reg] : NEG_INF;
          float w3 = (colb + 48 < N) ? acc[m][3][reg] : NEG_INF;
          float gmax = fmaxf(fmaxf(w0, w1), fmaxf(w2, w3));
#pragma unroll
          for (int off = 1; off < 16; off <<= 1)
            gmax = fmaxf(gmax, __shfl_xor(gmax, off, 64));
          if (gmax > rwall) qmask |= 1u << (m * 4 + reg);
        }
      }
    } else {
      const int colb = col0 + wc * 64 + cl;
#pragma unroll
      for (int m = 0; m < 4; ++m) {
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int row = wr * 64 + m * 16 + g * 4 + reg;
          const int lbase = (wc * BM + row) * KMAX;
          // EPI_MODE 8 reads the per-row threshold from the register
          // cache (one cross-lane shuffle) instead of a volatile LDS
          // read; the 16 unrolled call-boundary reads otherwise cost
          // ~70 serialized lgkmcnt(0) waits per tile per wave.
          float rwarm, rmin0;
          if constexpr (EPI_MODE >= 8 && EPI_MODE <= 11) {
            rwarm = __shfl(rmin_reg, m * 16 + g * 4 + reg, 64);
            rmin0 = rwarm;
          } else {
            rwarm = __shfl(warm, m * 16 + g * 4 + reg, 64);
            rmin0 = fmaxf(lsc[lbase], rwarm);
          }
          float w0 = (colb + 0 < N) ? acc[m][0][reg] : NEG_INF;
          float w1 = (colb + 16 < N) ? acc[m][1][reg] : NEG_INF;
          float w2 = (colb + 32 < N) ? acc[m][2][reg] : NEG_INF;
          float w3 = (colb + 48 < N) ? acc[m][3][reg] : NEG_INF;
          // EPI_MODE 11: one ballot replaces the 4-deep shfl_xor max
          // reduce in the pre-check (same predicate: does ANY lane of
          // the group beat the threshold) — halves the cross-lane
          // dependency chain of the hot no-candidate path.
          bool qual;
          if constexpr (EPI_MODE == 11 || EPI_MODE == 12) {
            const float lmax = fmaxf(fmaxf(w0, w1), fmaxf(w2, w3));
            qual = (__ballot(lmax > rmin0) & (0xFFFFull << (g * 16))) != 0;
          } else {
            float gmax = fmaxf(fmaxf(w0, w1), fmaxf(w2, w3));
#pragma unroll
            for (int off = 1; off < 16; off <<= 1)
              gmax = fmaxf(gmax, __shfl_xor(gmax, off, 64));
            qual = gmax > rmin0;
          }
          if (qual) {
            if constexpr (EPI_MODE == 3) {
              if (cl == 0 && stats) atomicAdd(&stats[0], 1ull);
            }
            if constexpr (EPI_MODE == 4) {
              // k=1: group argmax, leader max-merges into list slot 0
              float b = w0;
              int bn = 0;
              if (w1 > b) { b = w1; bn = 1; }
              if (w2 > b) { b = w2; bn = 2; }
              if (w3 > b) { b = w3; bn = 3; }
              float mv = b;
              int mlane = lane;
#pragma unroll
              for (int off = 1; off < 16; off <<= 1) {
                const float ov = __shfl_xor(mv, off, 64);
                const int ol = __shfl_xor(mlane, off, 64);
                if (ov > mv || (ov == mv && ol < mlane)) { mv = ov; mlane = ol; }
              }
              if (lane == mlane && mv > lsc[lbase]) {
                lsc[lbase] = mv;
                lix[lbase] = colb + bn * 16;
              }
            } else if constexpr (EPI_MODE == 6) {
              topk_extract_group_bl<true>(lsc, lix, lbase, rwarm, w0, w1,
                                          w2, w3, colb, N, lane, g, rowthr,
                                          (row0 + row < B) ? row0 + row + 1 : 0);
            } else if constexpr (EPI_MODE >= 9 && EPI_MODE <= 12) {
              // EPI_MODE 10: during list bootstrap (no threshold yet)
              // the inline insert would almost always fall through to
              // the callee anyway — skip straight to it. (A flag, not a
              // goto: a goto would block loop unrolling and trip the
              // runtime-indexed-accumulator scratch path.)
              bool storm = false;
              if constexpr (EPI_MODE == 10) {
                if (rwarm <= NEG_INF) {
                  storm = true;
                  topk_extract_group<true>(lsc, lix, lbase, rwarm, w0, w1,
                                           w2, w3, colb, N, lane, g, rowthr,
                                           (row0 + row < B) ? row0 + row + 1
                                                            : 0);
                  qmask = 1;
                }
              }
              if (!storm) {
              // inline single-insert fast path: the group argmax leader
              // inserts directly (no call, so no callee-entry
              // s_waitcnt vmcnt(0) draining the staging stream); only
              // groups with a SECOND candidate above the new min take
              // the noinline extractor.
              float b = w0;
              int bn = 0;
              if (w1 > b) { b = w1; bn = 1; }
              if (w2 > b) { b = w2; bn = 2; }
              if (w3 > b) { b = w3; bn = 3; }
              float mv = b;
              int mlane = lane;
#pragma unroll
              for (int off = 1; off < 16; off <<= 1) {
                const float ov = __shfl_xor(mv, off, 64);
                const int ol = __shfl_xor(mlane, off, 64);
                if (ov > mv || (ov == mv && ol < mlane)) { mv = ov; mlane = ol; }
              }
              float nmn = mv;  // new list min (valid on the leader lane)
              if (lane == mlane) {
                int nmp = 0;
#pragma unroll
                for (int q = 1; q < KMAX; ++q) {
                  const float sq = lsc[lbase + q];
                  if (sq < nmn) { nmn = sq; nmp = q; }
                }
                const int gcol = colb + bn * 16;
                if (nmp == 0) {
                  lsc[lbase] = mv;
                  lix[lbase] = gcol;
                } else {
                  const int mi2 = lix[lbase + nmp];
                  lsc[lbase] = nmn;
                  lix[lbase] = mi2;
                  lsc[lbase + nmp] = mv;
                  lix[lbase + nmp] = gcol;
                }
                if (bn == 0) w0 = NEG_INF;
                else if (bn == 1) w1 = NEG_INF;
                else if (bn == 2) w2 = NEG_INF;
                else w3 = NEG_INF;
                if (rowthr != nullptr && nmn > NEG_INF && row0 + row < B)
                  atomicMax(&rowthr[row0 + row], enc_f32(nmn));
              }
              qmask = 1;
              // does any remaining candidate still beat the new min?
              const float nmb = fmaxf(__shfl(nmn, mlane, 64), rwarm);
              const float g2l = fmaxf(fmaxf(w0, w1), fmaxf(w2, w3));
              bool more;
              if constexpr (EPI_MODE == 11 || EPI_MODE == 12) {
                more = (__ballot(g2l > nmb) & (0xFFFFull << (g * 16))) != 0;
              } else {
                float g2 = g2l;
#pragma unroll
                for (int off = 1; off < 16; off <<= 1)
                  g2 = fmaxf(g2, __shfl_xor(g2, off, 64));
                more = g2 > nmb;
              }
              if (more)
                topk_extract_group<true>(lsc, lix, lbase, rwarm, w0, w1, w2,
                                         w3, colb, N, lane, g, rowthr,
                                         (row0 + row < B) ? row0 + row + 1 : 0);
              }
            } else if constexpr (EPI_MODE != 2) {
              topk_extract_group<true>(lsc, lix, lbase, rwarm, w0, w1, w2,
                                       w3, colb, N, lane, g, rowthr,
                                       (row0 + row < B) ? row0 + row + 1 : 0);
              if constexpr (EPI_MODE == 8) qmask = 1;
            }
          }
        }
      }
      // refresh the register threshold cache from the settled list minima
      // (wave-local: one coalesced LDS read per lane, before the barrier)
      if constexpr (EPI_MODE >= 8 && EPI_MODE <= 12) {
        if (__any(qmask != 0))
          rmin_reg = fmaxf(rmin_reg, lsc[(wc * BM + wr * 64 + lane) * KMAX]);
      }
    }
    __syncthreads();  // next tile's first K-tile staged; lists settled
    cur ^= 1;

    if constexpr (EPI_MODE == 7) {
      // deferred extraction: prefetch for tile j+1 has completed at the
      // barrier above, so the callee's entry s_waitcnt vmcnt(0) is free
      // and the glds stream it would otherwise drain is untouched.
      if (__any(qmask != 0)) {
        const int colb = col0 + wc * 64 + cl;
#pragma unroll
        for (int m = 0; m < 4; ++m) {
#pragma unroll
          for (int reg = 0; reg < 4; ++reg) {
            if (qmask & (1u << (m * 4 + reg))) {
              const int row = wr * 64 + m * 16 + g * 4 + reg;
              const int lbase = (wc * BM + row) * KMAX;
              const float rwall = __shfl(rmin_reg, m * 16 + g * 4 + reg, 64);
              float w0 = (colb + 0 < N) ? acc[m][0][reg] : NEG_INF;
              float w1 = (colb + 16 < N) ? acc[m][1][reg] : NEG_INF;
              float w2 = (colb + 32 < N) ? acc[m][2][reg] : NEG_INF;
              float w3 = (colb + 48 < N) ? acc[m][3][reg] : NEG_INF;
              topk_extract_group<true>(lsc, lix, lbase, rwall, w0, w1, w2,
                                       w3, colb, N, lane, g, rowthr,
                                       (row0 + row < B) ? row0 + row + 1 : 0);
            }
          }
        }
        // refresh the register threshold from the settled list minima
        // (slot 0 is each list's min): one coalesced LDS read per lane.
        rmin_reg = fmaxf(rmin_reg, lsc[(wc * BM + wr * 64 + lane) * KMAX]);
      }
    }
  }

  if constexpr (EPI_MODE == 7) __syncthreads();  // deferred lists settled

  // write partials: [B][nchunks][KMAX]; merge the two col-half lists
  if (tid < BM) {
    const int grow = row0 + tid;
    if (grow < B) {
      const size_t base = ((size_t)grow * nchunks + chunk_id) * KMAX;
      float fs[KMAX];
      int fi[KMAX];
#pragma unroll
      for (int q = 0; q < KMAX; ++q) {
        fs[q] = lsc[tid * KMAX + q];
        fi[q] = lix[tid * KMAX + q];
      }
#pragma unroll
      for (int q = 0; q < KMAX; ++q) {
        const float s = lsc[(BM + tid) * KMAX + q];
        const int ix = lix[(BM + tid) * KMAX + q];
        int mp = 0;
        float mn = fs[0];
#pragma unroll
        for (int r = 1; r < KMAX; ++r)
          if (fs[r] < mn) { mn = fs[r]; mp = r; }
        if (s > mn) {
#pragma unroll
          for (int r = 0; r < KMAX; ++r)
            if (r == mp) { fs[r] = s; fi[r] = ix; }
        }
      }
#pragma unroll
      for (int q = 0; q < KMAX; ++q) {
        partial_score[base + q] = fs[q];
        partial_idx[base + q] = fi[q];
      }
    }
  }
}

inline constexpr auto cosine_topk_partial = cosine_topk_partial_t<0, 2>;


// ===========================================================================
// 256x256-tile 8-phase counted-pipeline kernel (guide T3+T4+T5).
//
// Geometry: 512 threads = 8 waves as 2 row-halves (wr) x 4 col-quads (wc);
// output tile 256 queries x 256 corpus rows, per-wave 128x64 (acc 8x4
// fragments of 16x16). K advances in BK=64 "windows" of 4 phases; each
// phase computes one row-quadrant (2 m-frags x 4 n x 2 kk = 16 MFMA). The
// B fragments are read ONCE per window (phase 0) and stay in registers,
// which is what frees the B image bytes early enough that the NEXT
// window's staging can target the LIVE buffer:
//
//   staging schedule (per wave, reader-aligned 1 KiB pieces):
//     window t, ph0: A(t+1) pieces 0,1      -> buf[(t+1)&1]  (unread now)
//     window t, ph1: A(t+1) 2,3 + B(t+2) 0,1-> B into buf[t&1]: its B bytes
//     window t, ph2: B(t+2) pieces 2,3         were consumed at ph0
//   certification: one `s_waitcnt vmcnt(4)` at ph3 (before the barrier)
//   leaves exactly B(t+2)'s 4 glds in flight and proves A(t+1) + B(t+1)
//   (and older) landed -> the pipeline NEVER drains to vmcnt(0).
//
// Reader-aligned staging makes per-wave vmcnt certification sound: a
// wave stages exactly the quarter of each half-image that it (or its
// barrier-synchronised co-reader) consumes. Two raw barriers per phase:
// the second separates every wave's ds_read retirement (compiler lgkm
// before its MFMAs) from the next phase's glds landing on those bytes.
//
// LDS: 2 buffers x (A[2 halves] + B[2 halves]) x 16 KiB = 128 KiB
// + shared top-k lists 16 KiB = 144 KiB -> 1 block/CU, 2 waves/SIMD.
// ===========================================================================


constexpr int BM8 = 256;
constexpr int BN8 = 256;
constexpr int THREADS8 = 512;
constexpr int HALF8 = 16384;  // one [128][64] bf16 half-image
constexpr int EGCAP = 8;          // emission stash slots per (block, wave)
constexpr int ESTASH_STRIDE = 64 + EGCAP * 1024;  // meta + slots, bytes



// EPI_MODE: 0 = stash+drain epilogue, 1 = GEMM only, 5 = slab-deferred
// epilogue (acc quadrants stream to a per-block global slab at tile end;
// the NEXT tile's windows drain one 32-col slice each, so the top-k
// maintenance hides under the MFMA pipeline instead of being exposed
// after the last window), 6 = ballot-skip stash (a register ballot
// pre-check against the rowthr floors marks which waves have ANY
// qualifying candidate this tile; unflagged waves' stash+drain phases
// are skipped uniformly — best 8p variant: 686 TF vs 420 for mode 0),
// 7 = per-row LDS candidate queue + single drain with the stash loop
// kept for bootstrap/overflow repair (correct but spills ~490 B/lane:
// the accumulators must stay live across the repair path — see
// profiles/knn_kernel_history.md for why a queue-only version was
// abandoned at the 256-VGPR/2-wave cap).
// Rare-path drain for the emission epilogue (EPI_MODE 9): the cold path
// stashes qualifying groups' accumulator quads into the wave's private
// slice of the dead A-image (plain LDS stores, no calls while the 128
// accumulator VGPRs are live), then makes ONE call here per wave per
// cold tile. By call time the accumulators are dead, so the call costs
// no spills — earlier structures (per-group callee: 32-reg/tile
// pre-spill; inlined appends: 0.5-1 KiB/lane scratch) all lost to the
// acc-liveness-across-append problem. __noinline__ keeps the body out
// of the sweep's register allocation entirely.
// v3-era per-group append callee, kept for A/B isolation (EPI_MODE 14):
// calls it once per qualifying group with the accumulators still live.
__device__ __noinline__ void emit_candidates(
    unsigned long long* __restrict__ cand, unsigned* __restrict__ ccount,
    long ccap, int grow, float v0, float v1, float v2, float v3, int colb,
    int N, float thr) {
  const bool q0 = v0 >= thr && colb < N;
  const bool q1 = v1 >= thr && colb + 16 < N;
  const bool q2 = v2 >= thr && colb + 32 < N;
  const bool q3 = v3 >= thr && colb + 48 < N;
  const int myc = (int)q0 + (int)q1 + (int)q2 + (int)q3;
  if (!myc) return;
  unsigned pos = atomicAdd(&ccount[grow], (unsigned)myc);
  unsigned long long* crow = cand + (size_t)grow * ccap;
#pragma unroll
  for (int n = 0; n < 4; ++n) {
    const bool qn = n == 0 ? q0 : n == 1 ? q1 : n == 2 ? q2 : q3;
    const float vn = n == 0 ? v0 : n == 1 ? v1 : n == 2 ? v2 : v3;
    if (qn) {
      if (pos < (unsigned)ccap)
        crow[pos] = ((unsigned long long)enc_f32(vn) << 32) |
                    (unsigned)(0x7fffffff - (colb + n * 16));
      ++pos;
    }
  }
}

__device__ __noinline__ void emit_stashed(
    unsigned long long* __restrict__ cand, unsigned* __restrict__ ccount,
    long ccap, char* stash, int ng, int rowbase, int colb, int N, int B,
    float thr0, float thr1) {
  const int lane = threadIdx.x & 63;
  const int g = lane >> 4;
  for (int i = 0; i < ng; ++i) {
    const int gid = (int)((volatile unsigned*)stash)[i];
    const int m = gid >> 2, reg = gid & 3;
    const int rl = m * 16 + g * 4 + reg;
    const float thr = __shfl(m >= 4 ? thr1 : thr0, rl & 63, 64);
    const f32x4 v = *(const f32x4*)(stash + 64 + i * 1024 + lane * 16);
    const int grow = rowbase + rl;
    const bool q0 = v[0] >= thr && colb < N;
    const bool q1 = v[1] >= thr && colb + 16 < N;
    const bool q2 = v[2] >= thr && colb + 32 < N;
    const bool q3 = v[3] >= thr && colb + 48 < N;
    const int myc = (int)q0 + (int)q1 + (int)q2 + (int)q3;
    if (!myc || grow >= B) continue;
    unsigned pos = atomicAdd(&ccount[grow], (unsigned)myc);
    unsigned long long* crow = cand + (size_t)grow * ccap;
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      const bool qn = n == 0 ? q0 : n == 1 ? q1 : n == 2 ? q2 : q3;
      if (qn) {
        if (pos < (unsigned)ccap)
          crow[pos] = ((unsigned long long)enc_f32(v[n]) << 32) |
                      (unsigned)(0x7fffffff - (colb + n * 16));
        ++pos;
      }
    }
  }
}

template <int EPI_MODE>
__global__ __launch_bounds__(THREADS8, 2) void cosine_topk_partial8p_t(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ C,
    float* __restrict__ partial_score, int* __restrict__ partial_idx,
    int B, int N, int D, int chunk_tiles, int nchunks,
    unsigned* rowthr = nullptr, unsigned long long* stats = nullptr,
    float* __restrict__ slab = nullptr,
    unsigned long long* __restrict__ cand = nullptr,
    unsigned* __restrict__ ccount = nullptr, long ccap = 0,
    char* __restrict__ estash = nullptr) {
  __shared__ char smem[8 * HALF8 + 2 * BM8 * KMAX * 4 + 32 + BM8 * 4 + 32];
  char* const smem0 = smem;
  // buffer b in {0,1}: A half h at b*4*HALF8 + h*HALF8; B half h at +2*HALF8
  auto ahalf = [&](int b, int h) -> char* {
    return smem0 + (b * 4 + h) * HALF8;
  };
  auto bhalf = [&](int b, int h) -> char* {
    return smem0 + (b * 4 + 2 + h) * HALF8;
  };
  float* lsc = (float*)(smem + 8 * HALF8);
  int* lix = (int*)(smem + 8 * HALF8 + BM8 * KMAX * 4);
  int* wflags = (int*)(smem + 8 * HALF8 + 2 * BM8 * KMAX * 4);
  int* ccnt = (int*)(smem + 8 * HALF8 + 2 * BM8 * KMAX * 4 + 32);
  unsigned* covf =
      (unsigned*)(smem + 8 * HALF8 + 2 * BM8 * KMAX * 4 + 32 + BM8 * 4);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int g = lane >> 4;
  const int cl = lane & 15;
  const int wr = wid >> 2;  // row half
  const int wc = wid & 3;   // col quad

  const int nrt = gridDim.y;
  int chunk_id, row_tile;
  if ((gridDim.x & 7) == 0 && gridDim.x * nrt >= 256) {
    const int bid = blockIdx.x + gridDim.x * blockIdx.y;
    const int xcd = bid & 7;
    const int slot = bid >> 3;
    const int cpx = gridDim.x >> 3;
    chunk_id = xcd * cpx + slot / nrt;
    row_tile = slot % nrt;
  } else {
    chunk_id = blockIdx.x;
    row_tile = blockIdx.y;
  }

  const int row0 = row_tile * BM8;
  const long rb = (long)D * 2;
  const int ntiles_total = (N + BN8 - 1) / BN8;
  const int tile0 = chunk_id * chunk_tiles;
  const int tiles_here = min(chunk_tiles, ntiles_total - tile0);
  const int nkt = D / 64;  // windows per col tile

  for (int i = tid; i < BM8 * KMAX; i += THREADS8) {
    lsc[i] = NEG_INF;
    lix[i] = -1;
  }
  __syncthreads();
  if (tiles_here <= 0) {
    // padded chunk: emit -inf partials and exit
    if (tid < BM8 && row0 + tid < B) {
      const size_t base = ((size_t)(row0 + tid) * nchunks + chunk_id) * KMAX;
#pragma unroll
      for (int q = 0; q < KMAX; ++q) {
        partial_score[base + q] = NEG_INF;
        partial_idx[base + q] = -1;
      }
    }
    return;
  }

  // ---- staging helpers --------------------------------------------------
  // A-share of K-tile t: quarter wc (rows wc*32..+32) of A-half(wr).
  // B-share of K-tile t: rows (wc&1)*64 + wr*32 ..+32 of B-half(wc>>1).
  // piece = 1 KiB = 8 image rows; lane-linear glds dest; swizzled source.
  auto stage_a_piece = [&](int t, int i) {  // i in 0..3
    const int kt = t % nkt;
    const int jt = t / nkt;
    if (jt >= tiles_here) return;
    char* img = ahalf(t & 1, wr);
    const int off = wc * 4096 + i * 1024;
    const int P = off + lane * 16;
    const int r = P >> 7;
    const int s_log = ((P >> 4) & 7) ^ (r & 7);
    const int gr = min(row0 + wr * 128 + r, B - 1);
    glds16((const char*)Q + (size_t)gr * rb + kt * 128 + s_log * 16, img + off);
  };
  auto stage_b_piece = [&](int t, int i) {
    const int kt = t % nkt;
    const int jt = t / nkt;
    if (jt >= tiles_here) return;
    char* img = bhalf(t & 1, wc >> 1);
    const int off = ((wc & 1) * 64 + wr * 32) * 128 + i * 1024;
    const int P = off + lane * 16;
    const int r = P >> 7;
    const int s_log = ((P >> 4) & 7) ^ (r & 7);
    const int col_base = (tile0 + jt) * BN8 + (wc >> 1) * 128;
    const int gr = min(col_base + r, N - 1);
    glds16((const char*)C + (size_t)gr * rb + kt * 128 + s_log * 16, img + off);
  };

  // ---- prologue: K-tile 0 fully + B(1); then steady schedule ------------
#pragma unroll
  for (int i = 0; i < 4; ++i) stage_a_piece(0, i);
#pragma unroll
  for (int i = 0; i < 4; ++i) stage_b_piece(0, i);
#pragma unroll
  for (int i = 0; i < 4; ++i) stage_b_piece(1, i);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  const int total_windows = tiles_here * nkt;

  // EPI_MODE 15: block-lifetime per-lane threshold VECTORS. Under the
  // emission main launch rowthr is static (prepass-published floors; mode
  // 9/15 never write it), so the 32 per-(m,reg) thresholds each lane
  // compares against are constants for the whole block. Mode 9 re-derives
  // them with 32 ds_bpermute + 32 vcc-serialised ballots per TILE; here
  // they are shuffled ONCE into 16 registers as bf16 PAIRS (a full-f32
  // tv[32] spilled 32 B/lane at the 256-VGPR/2-wave cap), rounded toward
  // -inf so the packed floor is <= the exact one. The per-tile hot sweep
  // becomes a pure-VALU running max-diff with a single ballot.
  // Exactness: qualification vs the rounded-DOWN floor is a superset of
  // qualification vs the exact floor, and a hot-sweep positive only opens
  // the cold path, whose per-lane compares in emit_stashed re-check the
  // EXACT thr0/thr1 before any candidate is stored.
  unsigned tvp[16];
  if constexpr (EPI_MODE == 15) {
    const int r0g = row0 + wr * 128 + lane;
    const float t0b =
        (r0g < B) ? (rowthr ? dec_f32(rowthr[r0g]) : NEG_INF) : 1e38f;
    const float t1b = (r0g + 64 < B)
                          ? (rowthr ? dec_f32(rowthr[r0g + 64]) : NEG_INF)
                          : 1e38f;
    // upper-16 f32 bits rounded toward -inf (truncation rounds toward
    // zero, i.e. UP for negatives -> push one ulp down there; a
    // most-negative-finite pushed to 0xff80 = -inf still compares
    // conservatively)
    auto bfloor = [](float f) -> unsigned {
      const unsigned b = __float_as_uint(f);
      unsigned hi = b >> 16;
      if ((b & 0xffffu) && (b >> 31)) ++hi;
      return hi & 0xffffu;
    };
#pragma unroll
    for (int m = 0; m < 8; ++m)
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        const int rl0 = m * 16 + g * 4 + 2 * h;
        const unsigned lo = bfloor(__shfl(m >= 4 ? t1b : t0b, rl0 & 63, 64));
        const unsigned hi =
            bfloor(__shfl(m >= 4 ? t1b : t0b, (rl0 + 1) & 63, 64));
        tvp[m * 2 + h] = lo | (hi << 16);
      }
  }

  // slab-deferred epilogue state (EPI_MODE 5): this block's 256 KiB slab.
  // Thread tid<256 owns list row tid; its warm threshold is cached in a
  // register per tile. The drain is array-free (slot 0 of a list always
  // holds its min, so the running threshold is one LDS read) to keep the
  // in-window register footprint tiny next to the 128-reg accumulator.
  float* myslab = nullptr;
  if constexpr (EPI_MODE == 5) {
    const long bid_flat = blockIdx.x + (long)gridDim.x * blockIdx.y;
    myslab = slab + bid_flat * (BM8 * BN8);
  }
  auto drain_slice = [&](int prev_col0, int w, float warm_r) {
    if (tid >= BM8) return;
    const int row = tid;
    volatile float* lrow = lsc + row * KMAX;
    volatile int* irow = lix + row * KMAX;
    float rmin = fmaxf(lrow[0], warm_r);  // slot 0 is the list min
    const float* srow = myslab + row * BN8 + w * 32;
    bool dirty = false;
    float mn_out = 0.f;
#pragma unroll 2
    for (int ii = 0; ii < 8; ++ii) {
      const float4 v4 = *(const float4*)(srow + 4 * ii);
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const float v = e == 0 ? v4.x : e == 1 ? v4.y : e == 2 ? v4.z : v4.w;
        const int gc = prev_col0 + w * 32 + 4 * ii + e;
        if (v > rmin && gc < N) {
          // replace slot 0 (the min) with v, then restore the invariant
          float mn2 = 1e38f;
          int m2 = 0;
#pragma unroll
          for (int q = 1; q < KMAX; ++q) {
            const float s = lrow[q];
            if (s < mn2) { mn2 = s; m2 = q; }
          }
          if (v <= mn2) {
            lrow[0] = v;
            irow[0] = gc;
          } else {
            lrow[0] = mn2;
            irow[0] = irow[m2];
            lrow[m2] = v;
            irow[m2] = gc;
          }
          const float nmn = fminf(mn2, v);
          rmin = fmaxf(rmin, nmn);
          mn_out = nmn;
          dirty = true;
        }
      }
    }
    if (dirty && rowthr != nullptr && mn_out > NEG_INF && row0 + row < B)
      atomicMax(&rowthr[row0 + row], enc_f32(mn_out));
  };

  for (int j = 0; j < tiles_here; ++j) {
    const int col0 = (tile0 + j) * BN8;

    float warm5 = NEG_INF;
    if constexpr (EPI_MODE == 5) {
      if (rowthr != nullptr && tid < BM8 && row0 + tid < B)
        warm5 = dec_f32(rowthr[row0 + tid]);
    }
    // EPI_MODE 6/7/9: per-lane threshold floors for this wave-half's two
    // rows (rows beyond B get +inf so clamped-row garbage never flags)
    float thr0 = NEG_INF, thr1 = NEG_INF;
    if constexpr (EPI_MODE == 6 || EPI_MODE == 7 || EPI_MODE == 9 ||
                  EPI_MODE == 12 || EPI_MODE == 14 || EPI_MODE == 15) {
      const int r0g = row0 + wr * 128 + lane;
      thr0 = (r0g < B) ? (rowthr ? dec_f32(rowthr[r0g]) : NEG_INF) : 1e38f;
      thr1 = (r0g + 64 < B) ? (rowthr ? dec_f32(rowthr[r0g + 64]) : NEG_INF)
                            : 1e38f;
    }

    f32x4 acc[8][4];
#pragma unroll
    for (int m = 0; m < 8; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n) acc[m][n] = f32x4{0.f, 0.f, 0.f, 0.f};

    for (int kt = 0; kt < nkt; ++kt) {
      const int t = j * nkt + kt;  // global window index
      const char* Ai = ahalf(t & 1, wr);
      const char* Bi = bhalf(t & 1, wc >> 1) + ((wc & 1) * 64) * 128;
      bf16x8 bf[4][2];
#pragma unroll
      for (int qm = 0; qm < 4; ++qm) {
        // ---- reads (before barrier; compiler inserts lgkm before MFMA)
        bf16x8 af[2][2];
#pragma unroll
        for (int mm = 0; mm < 2; ++mm)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            af[mm][kk] = read_frag(Ai, (qm * 2 + mm) * 16 + cl, kk * 4 + g);
        if (qm == 0) {
#pragma unroll
          for (int n = 0; n < 4; ++n)
#pragma unroll
            for (int kk = 0; kk < 2; ++kk)
              bf[n][kk] = read_frag(Bi, n * 16 + cl, kk * 4 + g);
        }
        // ---- staging issues (see schedule above)
        if (qm == 0 && t + 1 < total_windows) {
          stage_a_piece(t + 1, 0);
          stage_a_piece(t + 1, 1);
        } else if (qm == 1) {
          if (t + 1 < total_windows) {
            stage_a_piece(t + 1, 2);
            stage_a_piece(t + 1, 3);
          }
          if (t + 2 < total_windows) {
            stage_b_piece(t + 2, 0);
            stage_b_piece(t + 2, 1);
          }
        } else if (qm == 2 && t + 2 < total_windows) {
          stage_b_piece(t + 2, 2);
          stage_b_piece(t + 2, 3);
        } else if (qm == 3) {
          // certify window t+1's A and B. Steady state: the newest 4
          // outstanding glds are B(t+2)'s, so vmcnt(4) proves A(t+1) and
          // B(t+1) landed. At the chunk tail no B(t+2) was issued, so the
          // newest 4 would be A(t+1) itself -> drain fully there (rare).
          if (t + 2 < total_windows)
            asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
          else
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_barrier();
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int kk = 0; kk < 2; ++kk)
#pragma unroll
          for (int mm = 0; mm < 2; ++mm)
#pragma unroll
            for (int n = 0; n < 4; ++n)
              acc[qm * 2 + mm][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  af[mm][kk], bf[n][kk], acc[qm * 2 + mm][n], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
        __builtin_amdgcn_s_barrier();  // read-retirement vs next glds
      }
      if constexpr (EPI_MODE == 5) {
        // drain one slice of the previous tile's slab under this window's
        // pipeline (slab stores were separated by >= one window of
        // barriers; same workgroup -> L1-visible)
        if (j > 0 && kt < 8) drain_slice(col0 - BN8, kt, warm5);
      }
    }

    // ---- top-k epilogue ---------------------------------------------------
    if constexpr (EPI_MODE == 5) {
      // stream the final accumulators to the block's slab; the next tile's
      // windows (or the post-loop flush) merge them into the lists
#pragma unroll
      for (int m = 0; m < 8; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
#pragma unroll
          for (int reg = 0; reg < 4; ++reg)
            myslab[(wr * 128 + m * 16 + g * 4 + reg) * BN8 + wc * 64 +
                   n * 16 + cl] = acc[m][n][reg];
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");  // slab stores done
      __builtin_amdgcn_s_barrier();
    } else if constexpr (EPI_MODE == 1) {
#pragma unroll
      for (int m = 0; m < 8; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          asm volatile("" ::"v"(acc[m][n]));
    } else if constexpr (EPI_MODE == 9 || EPI_MODE == 12 || EPI_MODE == 14 ||
                         EPI_MODE == 15) {
      // ---- threshold-emission epilogue (round 2): no lists, no stash
      // phases, no barriers — the 1017 TF GEMM core's full accumulator
      // sweep is a register compare + rare global append. Exactness: the
      // rowthr floors are prepass-published 8-deep list minima, i.e. the
      // 8th-best of a score SUBSET, which lower-bounds the corpus
      // 8th-best and hence the true k-th for any k <= 8 — so emitting
      // every score >= floor provably captures the whole top-k. The
      // companion emit_merge_topk kernel reduces the (expected ~8N/s per
      // row) candidates; per-row counts above ccap flag a host fallback.
      // pack (order-encoded score << 32) | (0x7fffffff - col): one u64
      // max = higher score, then lower col (torch tie-break).
      // Two-phase sweep: the HOT pass is pure compare/shfl/ballot with
      // NO call sites, so the compiler keeps the 128 accumulator VGPRs
      // unspilled (a single-pass sweep with per-group calls pre-spilled
      // 32 regs per tile per wave -> ~27 GB of scratch traffic at the
      // 10M bench, the measured 917-vs-1017 TF gap). The COLD pass
      // (wave-uniform qmask != 0, ~12% of tiles per wave) re-derives the
      // thresholds and makes the rare __noinline__ emit_candidates
      // calls; its spills live in that cold block only.
      const int colb = col0 + wc * 64 + cl;
      unsigned qm32 = 0;
      if constexpr (EPI_MODE == 15) {
        // mode 15 hot sweep: no shuffles (tv precomputed per block), no
        // per-group ballots — one running max of (gmax - thr) and ONE
        // wave ballot. The per-group qualification bits are only needed
        // on the rare cold path, where they are re-derived with the
        // ballots mode 9 pays unconditionally.
        auto thr_lo = [&](int m, int reg) -> float {
          const unsigned pk = tvp[m * 2 + (reg >> 1)];
          return __uint_as_float((reg & 1) ? (pk & 0xffff0000u) : (pk << 16));
        };
        float run = NEG_INF;
#pragma unroll
        for (int m = 0; m < 8; ++m)
#pragma unroll
          for (int reg = 0; reg < 4; ++reg) {
            const float gmax = fmaxf(fmaxf(acc[m][0][reg], acc[m][1][reg]),
                                     fmaxf(acc[m][2][reg], acc[m][3][reg]));
            run = fmaxf(run, gmax - thr_lo(m, reg));
          }
        if (__builtin_expect(__ballot(run >= 0.f) != 0, 0)) {
#pragma unroll
          for (int m = 0; m < 8; ++m)
#pragma unroll
            for (int reg = 0; reg < 4; ++reg) {
              const float gmax = fmaxf(fmaxf(acc[m][0][reg], acc[m][1][reg]),
                                       fmaxf(acc[m][2][reg], acc[m][3][reg]));
              if (__ballot(gmax >= thr_lo(m, reg)))
                qm32 |= 1u << (m * 4 + reg);
            }
        }
      } else {
#pragma unroll
        for (int m = 0; m < 8; ++m) {
#pragma unroll
          for (int reg = 0; reg < 4; ++reg) {
            const int rl = m * 16 + g * 4 + reg;  // 0..127 within this half
            const float thr = __shfl(m >= 4 ? thr1 : thr0, rl & 63, 64);
            const float gmax = fmaxf(fmaxf(acc[m][0][reg], acc[m][1][reg]),
                                     fmaxf(acc[m][2][reg], acc[m][3][reg]));
            if (__ballot(gmax >= thr))  // uniform
              qm32 |= 1u << (m * 4 + reg);
          }
        }
      }
      if constexpr (EPI_MODE == 12) {
        // probe-only: hot sweep computed, cold path compiled out — lets
        // the host isolate the sweep's cost from the emission cost
        asm volatile("" ::"s"(qm32));
      } else if constexpr (EPI_MODE == 14) {
        // isolation A/B: the v3-era cold path (per-group callee with the
        // accumulators live; pays the pre-spill, no stash, no barrier)
        if (__builtin_expect(qm32 != 0, 0)) {
#pragma unroll
          for (int m = 0; m < 8; ++m) {
#pragma unroll
            for (int reg = 0; reg < 4; ++reg) {
              if (!(qm32 & (1u << (m * 4 + reg)))) continue;
              const int rl = m * 16 + g * 4 + reg;
              const float thr = __shfl(m >= 4 ? thr1 : thr0, rl & 63, 64);
              const float gmax = fmaxf(fmaxf(acc[m][0][reg], acc[m][1][reg]),
                                       fmaxf(acc[m][2][reg], acc[m][3][reg]));
              const int grow = row0 + wr * 128 + rl;
              if (gmax >= thr && grow < B)
                emit_candidates(cand, ccount, ccap, grow, acc[m][0][reg],
                                acc[m][1][reg], acc[m][2][reg],
                                acc[m][3][reg], colb, N, thr);
            }
          }
        }
      } else {
        if (__builtin_expect(qm32 != 0, 0)) {  // uniform cold path
          // stash qualifying groups into this (block, wave)'s private
          // slice of the GLOBAL emission scratch. An LDS stash in the
          // "dead" last-window A-image corrupted later tiles' scores
          // (isolated by the mode-14 A/B: v3 clean, LDS-stash dirty —
          // the image interacts with the cross-tile staging pipeline in
          // a way the stash-phase epilogues' barrier discipline tolerates
          // but a late per-wave write does not); global scratch has no
          // aliasing with the pipeline at all and needs no barrier.
          char* stash =
              estash +
              ((size_t)(blockIdx.x + (size_t)gridDim.x * blockIdx.y) * 8 +
               wid) *
                  ESTASH_STRIDE;
          int ng = 0;
#pragma unroll
          for (int m = 0; m < 8; ++m) {
#pragma unroll
            for (int reg = 0; reg < 4; ++reg) {
              if (!(qm32 & (1u << (m * 4 + reg)))) continue;
              if (__builtin_expect(ng == EGCAP, 0)) {
                // stash full. Correlated query batches (e.g. the bench's
                // template-generated signatures) make qualifying groups
                // BURST on "hot" corpus tiles — dozens of groups at once
                // — so overflow must drain-and-refill, not bail (a
                // poison->host-fallback here fired ~600x/row-tile on the
                // bench and doubled the step). This mid-sweep call keeps
                // some accumulators live (cold-block spills only).
                emit_stashed(cand, ccount, ccap, stash, ng,
                             row0 + wr * 128, colb, N, B, thr0, thr1);
                ng = 0;
              }
              *(f32x4*)(stash + 64 + ng * 1024 + (size_t)lane * 16) =
                  f32x4{acc[m][0][reg], acc[m][1][reg], acc[m][2][reg],
                        acc[m][3][reg]};
              if (lane == 0)
                ((volatile unsigned*)stash)[ng] = (unsigned)(m * 4 + reg);
              ++ng;
            }
          }
          if (ng > 0)
            emit_stashed(cand, ccount, ccap, stash, ng, row0 + wr * 128,
                         colb, N, B, thr0, thr1);
        }
      }
    } else {
      // ---- EPI_MODE 6/7 pre-check: one register ballot per (m,reg)
      // row group against the rowthr floors; waves with no qualifying
      // candidate skip their stash+drain phase below (uniformly — the
      // flags live in LDS so every wave takes the same branch).
      bool boot = true;        // stash phases process every row
      bool run_phases = true;  // whether the stash loop runs at all
      unsigned qual = 0;       // per-(m,reg) qualifying bits (my groups)
      (void)boot;
      (void)run_phases;
      if constexpr (EPI_MODE == 6 || EPI_MODE == 7) {
        if constexpr (EPI_MODE == 7) {
          for (int i = tid; i < BM8; i += THREADS8) ccnt[i] = 0;
          if (tid < 8) covf[tid] = 0;
        }
#pragma unroll
        for (int m = 0; m < 8; ++m) {
#pragma unroll
          for (int reg = 0; reg < 4; ++reg) {
            const int rl = m * 16 + g * 4 + reg;  // 0..127 within half
            const float thr = __shfl(m >= 4 ? thr1 : thr0, rl & 63, 64);
            const float lmax =
                fmaxf(fmaxf(acc[m][0][reg], acc[m][1][reg]),
                      fmaxf(acc[m][2][reg], acc[m][3][reg]));
            if (__ballot(lmax > thr) & (0xFFFFull << (g * 16)))
              qual |= 1u << (m * 4 + reg);
          }
        }
        if (lane == 0) wflags[wid] = __any(qual != 0) ? 1 : 0;
        __syncthreads();  // flags + queue counters visible
      }
      if constexpr (EPI_MODE == 7) {
        // EPI_MODE 7: one-pass LDS candidate queue. Qualifying lanes
        // append (score, col) to their row's slot array; one thread per
        // row folds them into the shared list. The 8-phase stash loop
        // runs only for bootstrap tiles (most waves flagged: thresholds
        // not yet established) and for rows whose queue overflowed —
        // steady state pays 3 barriers/tile instead of 16.
        constexpr int CQ8 = 8;
        const int t_last7 = j * nkt + nkt - 1;
        float* csc = (float*)ahalf(t_last7 & 1, 0);    // [256][CQ8]
        int* ccol = (int*)(csc + BM8 * CQ8);           // [256][CQ8]
        int nfl = 0;
#pragma unroll
        for (int i = 0; i < 8; ++i) nfl += wflags[i];
        boot = nfl >= 6;
        run_phases = boot;
        if (!boot) {
          if (qual) {
            const int colb = col0 + wc * 64 + cl;
#pragma unroll
            for (int m = 0; m < 8; ++m) {
#pragma unroll
              for (int reg = 0; reg < 4; ++reg) {
                if (qual & (1u << (m * 4 + reg))) {
                  const int row = wr * 128 + m * 16 + g * 4 + reg;
                  const float thr =
                      __shfl(m >= 4 ? thr1 : thr0,
                             (m * 16 + g * 4 + reg) & 63, 64);
#pragma unroll
                  for (int n = 0; n < 4; ++n) {
                    const float v = acc[m][n][reg];
                    const int gc = colb + n * 16;
                    if (v > thr && gc < N) {
                      const int pos = atomicAdd(&ccnt[row], 1);
                      if (pos < CQ8) {
                        csc[row * CQ8 + pos] = v;
                        ccol[row * CQ8 + pos] = gc;
                      } else {
                        atomicOr(&covf[row >> 5], 1u << (row & 31));
                      }
                    }
                  }
                }
              }
            }
          }
          __syncthreads();  // queue complete
          if (tid < BM8) {
            const int row = tid;
            const int grow = row0 + row;
            const bool ovf = (covf[row >> 5] >> (row & 31)) & 1;
            int nq = ccnt[row];
            if (nq > CQ8) nq = CQ8;
            if (!ovf && nq > 0 && grow < B) {
              // in-place LDS list update (slot 0 holds the min): no
              // register arrays — acc (128 VGPRs) must stay live for
              // the overflow stash pass, so the drain's footprint has
              // to be tiny to avoid scratch spill.
              volatile float* lrow = lsc + row * KMAX;
              volatile int* irow = lix + row * KMAX;
              float rmin = lrow[0];
              bool dirty = false;
              float mn_out = NEG_INF;
              for (int i = 0; i < nq; ++i) {
                const float v = csc[row * CQ8 + i];
                const int gc = ccol[row * CQ8 + i];
                if (v > rmin) {
                  float mn2 = 1e38f;
                  int m2 = 0;
#pragma unroll
                  for (int q = 1; q < KMAX; ++q) {
                    const float sq = lrow[q];
                    if (sq < mn2) { mn2 = sq; m2 = q; }
                  }
                  if (v <= mn2) {
                    lrow[0] = v;
                    irow[0] = gc;
                  } else {
                    lrow[0] = mn2;
                    irow[0] = irow[m2];
                    lrow[m2] = v;
                    irow[m2] = gc;
                  }
                  const float nmn = fminf(mn2, v);
                  rmin = nmn;
                  mn_out = nmn;
                  dirty = true;
                }
              }
              if (dirty && rowthr != nullptr && mn_out > NEG_INF)
                atomicMax(&rowthr[grow], enc_f32(mn_out));
            }
          }
          __syncthreads();  // lists settled; covf stable
          run_phases = (covf[0] | covf[1] | covf[2] | covf[3] | covf[4] |
                        covf[5] | covf[6] | covf[7]) != 0;
        }
      }
      // ---- top-k epilogue (stash + lane-parallel register-list drain) --
      // The A images of the LAST window's buffer are dead during the
      // epilogue (A(t+2) staging is only issued at window t+1 phase 0),
      // so each wave stashes its 128x64 score quadrant there and drains
      // it with one lane per two rows: candidates merge into a
      // register-held copy of the row's list (select chains, no calls,
      // no ballots, no volatile round-trips). Col-quad phases serialise
      // writers of the shared per-row lists.
      const int t_last = j * nkt + nkt - 1;
      // 32 KiB ([128][64] f32) of LDS is free during the epilogue (the A
      // images of the last window's buffer). The 8 waves take turns
      // stashing their 128x64 score quadrant there; after each stash ALL
      // waves drain it cooperatively (16 rows per wave, one row per lane,
      // single-writer lists), so the scan latency is hidden by 8-wave TLP
      // instead of being exposed on a lone wave.
      float* stash = (float*)ahalf(t_last & 1, 0);
#pragma unroll
      for (int phw = 0; phw < 8; ++phw) {
        if constexpr (EPI_MODE == 7) {
          if (!run_phases) break;  // uniform (LDS-derived)
        }
        if constexpr (EPI_MODE == 6 || EPI_MODE == 7) {
          // stashing wave's wid = swr*4 + swc; uniform (flags in LDS)
          if (!wflags[(phw & 1) * 4 + (phw >> 1)]) continue;
        }
        const int swc = phw >> 1, swr = phw & 1;  // stashing wave
        if (wc == swc && wr == swr) {
#pragma unroll
          for (int m = 0; m < 8; ++m)
#pragma unroll
            for (int n = 0; n < 4; ++n)
#pragma unroll
              for (int reg = 0; reg < 4; ++reg)
                stash[(m * 16 + g * 4 + reg) * 64 + n * 16 + cl] =
                    acc[m][n][reg];
        }
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();  // stash visible to every wave

        if constexpr (EPI_MODE != 2) {
          if (lane < 16) {  // 8 waves x 16 lanes = one drainer per row
            const int rl = wid * 16 + lane;         // 0..127 within half
            const int row = swr * 128 + rl;
            const int grow = row0 + row;
            // mode 7: outside bootstrap, the stash pass only repairs
            // rows whose candidate queue overflowed (queue-inserted
            // rows are already exact; re-scanning them would be
            // harmless but wasteful)
            bool skip_row = false;
            if constexpr (EPI_MODE == 7)
              skip_row = !boot && !((covf[row >> 5] >> (row & 31)) & 1);
            if (!skip_row) {
            const int colq = col0 + swc * 64;
            float ls[KMAX];
            int li[KMAX];
#pragma unroll
            for (int q = 0; q < KMAX; ++q) {
              ls[q] = lsc[row * KMAX + q];
              li[q] = lix[row * KMAX + q];
            }
            float rmin = ls[0];
#pragma unroll
            for (int q = 1; q < KMAX; ++q) rmin = fminf(rmin, ls[q]);
            if (rowthr != nullptr && grow < B)
              rmin = fmaxf(rmin, dec_f32(rowthr[grow]));
            bool dirty = false;
            const float* srow = stash + rl * 64;
#pragma unroll 4
            for (int ii = 0; ii < 16; ++ii) {
              const int iv = (ii + rl) & 15;  // bank stagger across rows
              const float4 v4 = *(const float4*)(srow + 4 * iv);
#pragma unroll
              for (int e = 0; e < 4; ++e) {
                const float v = e == 0 ? v4.x : e == 1 ? v4.y : e == 2 ? v4.z : v4.w;
                const int gc = colq + 4 * iv + e;
                if (v > rmin && gc < N) {
                  int mp = 0;
                  float mn1 = ls[0], mn2 = 1e38f;
#pragma unroll
                  for (int q = 1; q < KMAX; ++q) {
                    if (ls[q] < mn1) { mn2 = mn1; mn1 = ls[q]; mp = q; }
                    else if (ls[q] < mn2) { mn2 = ls[q]; }
                  }
#pragma unroll
                  for (int q = 0; q < KMAX; ++q)
                    if (q == mp) { ls[q] = v; li[q] = gc; }
                  rmin = fmaxf(rmin, fminf(mn2, v));
                  dirty = true;
                }
              }
            }
            if (dirty) {
              int mp = 0;
              float mn = ls[0];
#pragma unroll
              for (int q = 1; q < KMAX; ++q)
                if (ls[q] < mn) { mn = ls[q]; mp = q; }
              // swap the min into slot 0 (list invariant), then write back
              const float s_mp = ls[mp];
              const int i_mp = li[mp];
              ls[mp] = ls[0];
              li[mp] = li[0];
              ls[0] = s_mp;
              li[0] = i_mp;
#pragma unroll
              for (int q = 0; q < KMAX; ++q) {
                lsc[row * KMAX + q] = ls[q];
                lix[row * KMAX + q] = li[q];
              }
              if (rowthr != nullptr && mn > NEG_INF && grow < B)
                atomicMax(&rowthr[grow], enc_f32(mn));
            }
            }
          }
        }
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();  // lists settled; stash reusable
      }
      // drain the drains' rowthr publish atomics: they are vmcnt-tracked
      // and must not leak into the next tile's counted staging waits
      // (same audit class as the emission epilogue's vmcnt(0))
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
  }

  if constexpr (EPI_MODE == 5) {
    if (tiles_here > 0) {
      const int last_col0 = (tile0 + tiles_here - 1) * BN8;
      float warmf = NEG_INF;
      if (rowthr != nullptr && tid < BM8 && row0 + tid < B)
        warmf = dec_f32(rowthr[row0 + tid]);
      for (int w = 0; w < 8; ++w) drain_slice(last_col0, w, warmf);
    }
  }

  // write partials: [B][nchunks][KMAX] (emission modes have no lists —
  // their results went straight to the candidate buffer)
  if constexpr (EPI_MODE != 9 && EPI_MODE != 12 && EPI_MODE != 14) {
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    if (tid < BM8) {
      const int grow = row0 + tid;
      if (grow < B) {
        const size_t base = ((size_t)grow * nchunks + chunk_id) * KMAX;
#pragma unroll
        for (int q = 0; q < KMAX; ++q) {
          partial_score[base + q] = lsc[tid * KMAX + q];
          partial_idx[base + q] = lix[tid * KMAX + q];
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Publish the merged sample's k-th best as the per-row emission floor.
// After topk_merge over the prepass partials, out_s[row][kth] is the EXACT
// kth-best of the whole sampled column set — a far tighter (still exact)
// lower bound on the corpus kth-best than the per-(row, col-half) list
// minima the prepass publishes on its own (those are 8th-of-512-columns;
// at 10M rows they admitted ~40x more candidates and overflowed).
// ---------------------------------------------------------------------------
__global__ void publish_emission_floor(const float* __restrict__ merged_s,
                                       unsigned* __restrict__ rowthr, int B,
                                       int kth) {
  const int r = blockIdx.x * blockDim.x + threadIdx.x;
  if (r < B) {
    const float v = merged_s[(size_t)r * KMAX + kth];
    if (v > NEG_INF) atomicMax(&rowthr[r], enc_f32(v));
  }
}

// ---------------------------------------------------------------------------
// Small-batch streaming emission search (B <= 8): request-level serving.
//
// The MFMA kernels pad tiny query batches to a full 128/256-row tile, so
// a single /warn query pays 128x the GEMM compute (measured p50 6.4 ms
// at B=1 x 10M vs the ~2 ms corpus-stream bound). Here each THREAD
// streams whole corpus rows (64 lanes read 64 consecutive rows — every
// byte of HBM read exactly once, full-bandwidth pattern), keeps B <= 8
// fp32 accumulators in registers against the LDS-resident queries, and
// emits scores >= the per-query prepass floor into the same candidate
// buffer the emission path merges. Exactness = the emission argument.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void smallb_emit_kernel(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ C, int B,
    long N, int D, const unsigned* __restrict__ rowthr,
    unsigned long long* __restrict__ cand, unsigned* __restrict__ ccount,
    long ccap) {
  extern __shared__ char qmem[];  // [B][D] bf16 queries
  __shared__ float fl[8];
  for (int i = threadIdx.x; i < B * (D / 8); i += 256)
    ((bf16x8*)qmem)[i] = ((const bf16x8*)Q)[i];
  if (threadIdx.x < B) fl[threadIdx.x] = dec_f32(rowthr[threadIdx.x]);
  __syncthreads();

  const int nj = D / 8;  // D % 64 == 0 -> nj % 8 == 0
  for (long r = (long)blockIdx.x * 256 + threadIdx.x; r < N;
       r += (long)gridDim.x * 256) {
    // 4 independent partial accumulators per query: a single running
    // accumulator serializes a 768-deep FMA dependency chain per row
    // (measured 2.5x slower than the padded MFMA path); 4 chains + the
    // 4-vector load batch restore ILP and memory-level parallelism.
    f32x4 accv[8][2];
#pragma unroll
    for (int b = 0; b < 8; ++b)
      accv[b][0] = accv[b][1] = f32x4{0.f, 0.f, 0.f, 0.f};
    const bf16x8* row = (const bf16x8*)(C + r * D);
    for (int j = 0; j < nj; j += 8) {
      bf16x8 cv[8];
#pragma unroll
      for (int v = 0; v < 8; ++v) cv[v] = row[j + v];
#pragma unroll
      for (int b = 0; b < 8; ++b) {
        if (b < B) {
          const bf16x8* qb = (const bf16x8*)(qmem + (size_t)b * D * 2) + j;
#pragma unroll
          for (int v = 0; v < 8; ++v) {
            const bf16x8 qv = qb[v];
#pragma unroll
            for (int e = 0; e < 8; ++e)
              accv[b][v >> 2][v & 3] += (float)cv[v][e] * (float)qv[e];
          }
        }
      }
    }
    float acc[8];
#pragma unroll
    for (int b = 0; b < 8; ++b)
      acc[b] = ((accv[b][0][0] + accv[b][0][1]) +
                (accv[b][0][2] + accv[b][0][3])) +
               ((accv[b][1][0] + accv[b][1][1]) +
                (accv[b][1][2] + accv[b][1][3]));
#pragma unroll
    for (int b = 0; b < 8; ++b) {
      if (b < B && acc[b] >= fl[b]) {
        const unsigned pos = atomicAdd(&ccount[b], 1u);
        if (pos < (unsigned)ccap)
          cand[(size_t)b * ccap + pos] =
              ((unsigned long long)enc_f32(acc[b]) << 32) |
              (unsigned)(0x7fffffff - (int)r);
      }
    }
  }
}

// v4 lane remap (the serving DEFAULT; KAKVEDA_SMALLB=2 reverts): one
// wave covers 8 CONSECUTIVE rows with 8 lanes per row, so a single load
// instruction touches 8 FULL 128-byte lines, and scores are reduced
// across the 8 chunk lanes with 3 shfl_xor steps at row end; per-lane
// f32x4 sub-chains keep the FMA dependency depth at 2 per segment (the
// v1 lesson). Measured +13% over v2 at B=4 (parity at B=1), and TCC
// counters show WHY (profiles/pmc_round2.md): L2 traffic is identical
// (one request per 128-B line in both mappings — L1 coalesces v2's
// eight same-line 16-B loads), so the win is the leaner inner loop
// (1,587 vs 2,442 instructions) and 134-vs-186 VGPRs = 3-vs-2
// waves/SIMD occupancy, NOT cache behaviour. Exactness: identical
// emission contract to v2 — every score >= the shared prepass floor is
// emitted; tail rows are clamped for the load and guarded at emission.
__global__ __launch_bounds__(256) void smallb_emit_kernel_v4(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ C, int B,
    long N, int D, const unsigned* __restrict__ rowthr,
    unsigned long long* __restrict__ cand, unsigned* __restrict__ ccount,
    long ccap) {
  extern __shared__ char qmem[];  // [B][D] bf16 queries
  __shared__ float fl[8];
  for (int i = threadIdx.x; i < B * (D / 8); i += 256)
    ((bf16x8*)qmem)[i] = ((const bf16x8*)Q)[i];
  if (threadIdx.x < B) fl[threadIdx.x] = dec_f32(rowthr[threadIdx.x]);
  __syncthreads();

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int rr = lane >> 3;  // row within this wave's 8
  const int ch = lane & 7;   // 16-byte chunk within a 128-B segment
  const int nseg = D / 64;   // 128-B segments per row (D % 64 == 0)
  for (long r0 = (long)blockIdx.x * 32 + wid * 8; r0 < N;
       r0 += (long)gridDim.x * 32) {
    const long r = r0 + rr < N ? r0 + rr : N - 1;  // clamp tail loads
    const bf16x8* row = (const bf16x8*)(C + r * D);
    f32x4 accv[8];
#pragma unroll
    for (int b = 0; b < 8; ++b) accv[b] = f32x4{0.f, 0.f, 0.f, 0.f};
    int j = 0;
    for (; j + 4 <= nseg; j += 4) {
      bf16x8 cv[4];
#pragma unroll
      for (int v = 0; v < 4; ++v) cv[v] = row[(j + v) * 8 + ch];
#pragma unroll
      for (int b = 0; b < 8; ++b) {
        if (b < B) {
          const bf16x8* qb = (const bf16x8*)(qmem + (size_t)b * D * 2);
#pragma unroll
          for (int v = 0; v < 4; ++v) {
            const bf16x8 qv = qb[(j + v) * 8 + ch];
#pragma unroll
            for (int e = 0; e < 8; ++e)
              accv[b][e & 3] += (float)cv[v][e] * (float)qv[e];
          }
        }
      }
    }
    for (; j < nseg; ++j) {  // D % 256 != 0 tail segments
      const bf16x8 cv = row[j * 8 + ch];
#pragma unroll
      for (int b = 0; b < 8; ++b) {
        if (b < B) {
          const bf16x8 qv =
              ((const bf16x8*)(qmem + (size_t)b * D * 2))[j * 8 + ch];
#pragma unroll
          for (int e = 0; e < 8; ++e)
            accv[b][e & 3] += (float)cv[e] * (float)qv[e];
        }
      }
    }
#pragma unroll
    for (int b = 0; b < 8; ++b) {
      if (b < B) {
        float s = (accv[b][0] + accv[b][1]) + (accv[b][2] + accv[b][3]);
        s += __shfl_xor(s, 1, 64);
        s += __shfl_xor(s, 2, 64);
        s += __shfl_xor(s, 4, 64);  // the row group's 8 lanes now agree
        if (ch == 0 && r0 + rr < N && s >= fl[b]) {
          const unsigned pos = atomicAdd(&ccount[b], 1u);
          if (pos < (unsigned)ccap)
            cand[(size_t)b * ccap + pos] =
                ((unsigned long long)enc_f32(s) << 32) |
                (unsigned)(0x7fffffff - (int)(r0 + rr));
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Companion merge for the emission epilogue (8p EPI_MODE 9): exact top-k
// per row over the emitted (score, col) candidates. grid = B blocks x 256
// threads; each thread keeps a sorted top-KMAX of its strided slice in
// registers, then a log2(256)-step LDS tree merges them. Overflowed rows
// (count > cap: emission skipped stores) write idx[0] = -2 so the host
// falls back to the list-epilogue kernel — the exactness guard.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void emit_merge_topk(
    const unsigned long long* __restrict__ cand,
    const unsigned* __restrict__ ccount, float* __restrict__ out_s,
    long* __restrict__ out_i, int B, long cap, int k) {
  const int row = blockIdx.x;
  const int tid = threadIdx.x;
  const unsigned cnt = ccount[row];
  if (cnt > (unsigned)cap) {
    if (tid == 0) {
      for (int q = 0; q < k; ++q) {
        out_s[(size_t)row * k + q] = NEG_INF;
        out_i[(size_t)row * k + q] = q == 0 ? -2 : -1;
      }
    }
    return;
  }
  unsigned long long loc[KMAX];  // sorted desc; 0 = empty sentinel
#pragma unroll
  for (int q = 0; q < KMAX; ++q) loc[q] = 0ull;
  const unsigned long long* crow = cand + (size_t)row * cap;
  for (unsigned i = tid; i < cnt; i += 256) {
    const unsigned long long v = crow[i];
    if (v > loc[KMAX - 1]) {
      loc[KMAX - 1] = v;  // insert at tail, bubble up (array stays sorted)
#pragma unroll
      for (int q = KMAX - 1; q > 0; --q)
        if (loc[q] > loc[q - 1]) {
          const unsigned long long t = loc[q];
          loc[q] = loc[q - 1];
          loc[q - 1] = t;
        }
    }
  }
  __shared__ unsigned long long all[256 * KMAX];
#pragma unroll
  for (int q = 0; q < KMAX; ++q) all[tid * KMAX + q] = loc[q];
  __syncthreads();
  for (int stride = 128; stride >= 1; stride >>= 1) {
    if (tid < stride) {
      // two-pointer merge of two sorted-desc KMAX lists -> top KMAX
      // (LDS-indexed: a register array would go to scratch)
      const unsigned long long* pa = all + (size_t)tid * KMAX;
      const unsigned long long* pb = all + (size_t)(tid + stride) * KMAX;
      unsigned long long o[KMAX];
      int ia = 0, ib = 0;
#pragma unroll
      for (int q = 0; q < KMAX; ++q) {
        const unsigned long long va = pa[ia], vb = pb[ib];
        if (va >= vb) {
          o[q] = va;
          ++ia;
        } else {
          o[q] = vb;
          ++ib;
        }
      }
#pragma unroll
      for (int q = 0; q < KMAX; ++q) all[tid * KMAX + q] = o[q];
    }
    __syncthreads();
  }
  if (tid == 0) {
    for (int q = 0; q < k; ++q) {
      const unsigned long long v = all[q];
      if (v == 0ull) {
        out_s[(size_t)row * k + q] = NEG_INF;
        out_i[(size_t)row * k + q] = -1;
      } else {
        out_s[(size_t)row * k + q] = dec_f32((unsigned)(v >> 32));
        out_i[(size_t)row * k + q] = 0x7fffffffL - (long)(v & 0xffffffffu);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// k-means centroid update: segmented reduction of bf16 points into
// per-cluster fp32 sums + counts. Grid: (dim-tiles, point-chunks); each
// block accumulates a [C][64]-dim LDS partial over its point chunk, then
// adds it to the global sums once (C <= 256; LDS = C*64*4 <= 64 KiB).
// Points are read exactly once per dim-tile (coalesced 128-B row slices);
// bandwidth-bound by design, replacing torch index_add_ (which was ~30x
// over the I/O floor in benchmarks/kmeans_bench.py).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void kmeans_update_kernel(
    const bf16_t* __restrict__ points, const int* __restrict__ assign,
    float* __restrict__ sums, float* __restrict__ counts, int N, int D,
    int C, int points_per_chunk) {
  // C <= 128: FOUR replicas of the [C][64] partial, one per point lane, so
  // the inner loop is a plain LDS read+add+write on a slot owned by exactly
  // one thread — the single-replica per-element atomicAdd version measured
  // LDS-atomic-bound (~385 GB/s, 40 ms at 10M x 768). C > 128: one replica
  // with atomics (LDS budget).
  extern __shared__ float part[];  // [R][C][64] sums + [C] counts
  const int R = (C <= 128) ? 4 : 1;
  const int tid = threadIdx.x;
  const int d0 = blockIdx.x * 64;
  const int p0 = blockIdx.y * points_per_chunk;
  const int pend = min(p0 + points_per_chunk, N);

  float* cpart = part + R * C * 64;
  for (int i = tid; i < R * C * 64 + C; i += 256) part[i] = 0.f;
  __syncthreads();

  // thread (pl, d) = (tid>>6, tid&63); 16 points batched per pass so the
  // 2-byte strided loads stack up in flight (the 4-in-flight version was
  // load-latency-bound at ~350 GB/s: each iteration's LDS add depended on
  // its own just-issued load)
  const int pl = tid >> 6;
  const int d = tid & 63;
  float* mypart = part + (R == 4 ? pl * C * 64 : 0);
  constexpr int PU = 16;
  int p = p0 + pl;
  for (; p + 4 * (PU - 1) < pend; p += 4 * PU) {
    float v[PU];
    int a[PU];
#pragma unroll
    for (int u = 0; u < PU; ++u) {
      const int pp = p + 4 * u;
      a[u] = assign[pp];
      v[u] = (float)points[(size_t)pp * D + d0 + d];
    }
#pragma unroll
    for (int u = 0; u < PU; ++u) {
      if (a[u] >= 0 && a[u] < C) {
        if (R == 4)
          mypart[a[u] * 64 + d] += v[u];  // slot owned by this (pl, d) thread
        else
          atomicAdd(&part[a[u] * 64 + d], v[u]);
      }
    }
  }
  for (; p < pend; p += 4) {
    const int a = assign[p];
    const float v = (float)points[(size_t)p * D + d0 + d];
    if (a >= 0 && a < C) {
      if (R == 4)
        mypart[a * 64 + d] += v;
      else
        atomicAdd(&part[a * 64 + d], v);
    }
  }
  __syncthreads();

  for (int i = tid; i < C * 64; i += 256) {
    float v = part[i];
    if (R == 4) v += part[C * 64 + i] + part[2 * C * 64 + i] + part[3 * C * 64 + i];
    if (v != 0.f)
      atomicAdd(&sums[(size_t)(i / 64) * D + d0 + i % 64], v);
  }
  // counts: LDS partial first (the naive global-atomic version costs ~1M
  // contended RMWs per batch), then C adds per chunk
  if (blockIdx.x == 0) {
    for (int p = p0 + tid; p < pend; p += 256) {
      const int a = assign[p];
      if (a >= 0 && a < C) atomicAdd(&cpart[a], 1.0f);
    }
    __syncthreads();
    for (int i = tid; i < C; i += 256)
      if (cpart[i] != 0.f) atomicAdd(&counts[i], cpart[i]);
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
// Small-candidate merge: one THREAD per query row (the block-per-row merge
// above serialises a 2048-entry scan on thread 0, which dominates when B
// is large and nchunks small — e.g. k-means assignment at B=1M, C=64).
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void topk_merge_small(
    const float* __restrict__ partial_score, const int* __restrict__ partial_idx,
    float* __restrict__ out_score, long* __restrict__ out_idx, int B,
    int nchunks, int k) {
  const int row = blockIdx.x * blockDim.x + threadIdx.x;
  if (row >= B) return;
  const int total = nchunks * KMAX;
  const size_t base = (size_t)row * total;
  float ls[KMAX];
  int li[KMAX];
#pragma unroll
  for (int q = 0; q < KMAX; ++q) { ls[q] = NEG_INF; li[q] = -1; }
  float lmin = NEG_INF;
  for (int e = 0; e < total; ++e) {
    const float s = partial_score[base + e];
    if (s > lmin) {
      int mp = 0;
      float mn1 = ls[0], mn2 = 1e38f;
#pragma unroll
      for (int q = 1; q < KMAX; ++q) {
        if (ls[q] < mn1) { mn2 = mn1; mn1 = ls[q]; mp = q; }
        else if (ls[q] < mn2) { mn2 = ls[q]; }
      }
#pragma unroll
      for (int q = 0; q < KMAX; ++q)
        if (q == mp) { ls[q] = s; li[q] = partial_idx[base + e]; }
      lmin = fminf(mn2, s);
    }
  }
  // sort descending (insertion sort over KMAX)
#pragma unroll
  for (int a = 1; a < KMAX; ++a) {
    const float s = ls[a];
    const int ix = li[a];
    int b = a - 1;
    for (; b >= 0 && ls[b] < s; --b) { ls[b + 1] = ls[b]; li[b + 1] = li[b]; }
    ls[b + 1] = s;
    li[b + 1] = ix;
  }
  for (int q = 0; q < k; ++q) {
    out_score[(size_t)row * k + q] = ls[q] <= NEG_INF ? -INFINITY : ls[q];
    out_idx[(size_t)row * k + q] = li[q];
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

// ---------------------------------------------------------------------------
// k-means assignment: N points x C centroids (C <= 64), argmax per point.
//
// The general cosine_topk path is built for huge corpora: at the pattern-
// detector shape (N=10M points as "queries", C=64 centroids as the
// "corpus") its 12-window staged pipeline runs with one 64-valid-column
// tile, nothing to overlap, and a per-block B-restage — measured ~440 GB/s
// effective vs the ~8 TB/s point-stream bound (ROUND2.md). This kernel is
// shaped for that case instead:
//   - centroids are staged into LDS ONCE per block (row-padded pitch so the
//     16-lane fragment reads spread across all 64 banks) and stay resident
//     for every point tile the block processes;
//   - point MFMA A-fragments stream straight from HBM into registers
//     (16 B/lane, sector-coalesced) through a 1-window register prefetch —
//     no LDS staging, no per-window barriers at all;
//   - 8 waves x 32 rows = 256 points per tile, grid-strided so one launch
//     covers any N; the argmax epilogue is pure registers + shfl.
// ---------------------------------------------------------------------------
constexpr int ASSIGN_THREADS = 512;  // 8 waves

// Row pitch for the centroid LDS image: pad so pitch % 256 == 64, which
// makes the 64 lanes of a fragment read (16 rows x 4 slot-groups of 16 B)
// land on 64 distinct banks — the minimum 4 conflict-free phases for a
// 1 KiB ds_read_b128 wave. (A 16 B pad measured as a 4x conflict: lanes
// with equal cl+g collided.)
constexpr __host__ __device__ int assign_pitch(int D) {
  return D * 2 + ((64 - (D * 2) % 256) + 256) % 256;
}

__global__ __launch_bounds__(ASSIGN_THREADS, 1) void kmeans_assign_kernel(
    const bf16_t* __restrict__ P, const bf16_t* __restrict__ Cc,
    float* __restrict__ out_score, int* __restrict__ out_idx,
    int N, int D, int C) {
  extern __shared__ char cmem[];  // 64 centroid rows, padded pitch
  const int pitch = assign_pitch(D);
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int g = lane >> 4, cl = lane & 15;

  // stage centroids once; rows >= C zero-filled so their MFMAs stay finite
  for (int s = tid; s < 64 * (D / 8); s += ASSIGN_THREADS) {
    const int c = s / (D / 8);
    const int so = s % (D / 8);
    bf16x8 v = {};
    if (c < C)
      v = *(const bf16x8*)((const char*)Cc + ((size_t)c * D + (size_t)so * 8) * 2);
    *(bf16x8*)(cmem + (size_t)c * pitch + so * 16) = v;
  }
  __syncthreads();

  const long prow = (long)D * 2;
  const int nkt = D / 64;
  for (long tile0 = (long)blockIdx.x * 256; tile0 < N;
       tile0 += (long)gridDim.x * 256) {
    const long r0 = tile0 + (long)wid * 32;  // this wave's 32 rows
    f32x4 acc[2][4];
#pragma unroll
    for (int m = 0; m < 2; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n) acc[m][n] = f32x4{0.f, 0.f, 0.f, 0.f};

    // A fragments direct global->register. Straight-line, unroll-by-3
    // window loop with THREE named prefetch buffers and 2 windows always
    // in flight: a branchy 1-deep version compiled to a `s_waitcnt
    // vmcnt(0)` before every MFMA phase (the conditional prefetch block
    // defeats counted waits) and measured 153 GB/s; straight-line code
    // lets the compiler emit counted vmcnt waits so the point stream
    // pipelines. Tail rows clamp to the last point (results masked at
    // the write below); past-the-end windows clamp to the last window
    // (redundant re-reads, still straight-line).
    bf16x8 a0[2][2], a1[2][2], a2[2][2];  // [kk][m]
    auto lda = [&](bf16x8 (&dst)[2][2], int kt) {
      if (kt >= nkt) kt = nkt - 1;
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int m = 0; m < 2; ++m) {
          long r = r0 + m * 16 + cl;
          if (r >= N) r = N - 1;
          dst[kk][m] = *(const bf16x8*)((const char*)P + r * prow +
                                        (size_t)(kt * 8 + kk * 4 + g) * 16);
        }
    };
    auto mfma_win = [&](bf16x8 (&src)[2][2], int kt) {
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        bf16x8 bfrag[4];
        const int slot = kt * 8 + kk * 4 + g;
#pragma unroll
        for (int n = 0; n < 4; ++n)
          bfrag[n] =
              *(const bf16x8*)(cmem + (size_t)(n * 16 + cl) * pitch + slot * 16);
#pragma unroll
        for (int m = 0; m < 2; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n)
            acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                src[kk][m], bfrag[n], acc[m][n], 0, 0, 0);
      }
    };
    lda(a0, 0);
    lda(a1, 1);
    int kt = 0;
    for (; kt + 2 < nkt; kt += 3) {
      lda(a2, kt + 2);
      mfma_win(a0, kt);
      lda(a0, kt + 3);
      mfma_win(a1, kt + 1);
      lda(a1, kt + 4);
      mfma_win(a2, kt + 2);
    }
    if (kt < nkt) mfma_win(a0, kt);
    if (kt + 1 < nkt) mfma_win(a1, kt + 1);

    // argmax per row over 64 cols (cols >= C masked; ties -> lowest col,
    // matching torch.argmax / the CPU reference)
#pragma unroll
    for (int m = 0; m < 2; ++m) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float best = NEG_INF;
        int bcol = 0x7fffffff;
#pragma unroll
        for (int n = 0; n < 4; ++n) {
          const int col = n * 16 + cl;
          const float v = acc[m][n][reg];
          if (col < C && (v > best || (v == best && col < bcol))) {
            best = v;
            bcol = col;
          }
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1) {
          const float ov = __shfl_xor(best, off, 64);
          const int oc = __shfl_xor(bcol, off, 64);
          if (ov > best || (ov == best && oc < bcol)) {
            best = ov;
            bcol = oc;
          }
        }
        const long row = r0 + m * 16 + g * 4 + reg;
        if (cl == 0 && row < N) {
          out_score[row] = best;
          out_idx[row] = bcol;
        }
      }
    }
  }
}

}  // namespace kakveda

