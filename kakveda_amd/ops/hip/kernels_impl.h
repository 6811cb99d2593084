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

  for (int j = 0; j < tiles_here; ++j) {
    const int col0 = (tile0 + j) * BN;

    // per-lane warm thresholds: lane l caches rowthr for row wr*64+l; the
    // pre-check broadcasts the right lane's value with one shfl. Published
    // thresholds from other blocks prune chunk bootstraps to ~nothing.
    float warm = NEG_INF;
    if (EPI_MODE != 1 && rowthr != nullptr && row0 + wr * 64 + lane < B)
      warm = dec_f32(rowthr[row0 + wr * 64 + lane]);

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
#pragma unroll
        for (int m = 0; m < 4; ++m)
#pragma unroll
          for (int n = 0; n < 4; ++n)
            acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag[m], bfrag[n], acc[m][n], 0, 0, 0);
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
    if constexpr (EPI_MODE == 1) {
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          asm volatile("" ::"v"(acc[m][n]));
    } else {
      const int colb = col0 + wc * 64 + cl;
#pragma unroll
      for (int m = 0; m < 4; ++m) {
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int row = wr * 64 + m * 16 + g * 4 + reg;
          const int lbase = (wc * BM + row) * KMAX;
          const float rwarm = __shfl(warm, m * 16 + g * 4 + reg, 64);
          const float rmin0 = fmaxf(lsc[lbase], rwarm);
          float w0 = (colb + 0 < N) ? acc[m][0][reg] : NEG_INF;
          float w1 = (colb + 16 < N) ? acc[m][1][reg] : NEG_INF;
          float w2 = (colb + 32 < N) ? acc[m][2][reg] : NEG_INF;
          float w3 = (colb + 48 < N) ? acc[m][3][reg] : NEG_INF;
          float gmax = fmaxf(fmaxf(w0, w1), fmaxf(w2, w3));
#pragma unroll
          for (int off = 1; off < 16; off <<= 1)
            gmax = fmaxf(gmax, __shfl_xor(gmax, off, 64));
          if (gmax > rmin0) {
            if constexpr (EPI_MODE == 3) {
              if (cl == 0 && stats) atomicAdd(&stats[0], 1ull);
            }
            if constexpr (EPI_MODE == 6) {
              topk_extract_group_bl<true>(lsc, lix, lbase, rwarm, w0, w1,
                                          w2, w3, colb, N, lane, g, rowthr,
                                          (row0 + row < B) ? row0 + row + 1 : 0);
            } else if constexpr (EPI_MODE != 2) {
              topk_extract_group<true>(lsc, lix, lbase, rwarm, w0, w1, w2,
                                       w3, colb, N, lane, g, rowthr,
                                       (row0 + row < B) ? row0 + row + 1 : 0);
            }
          }
        }
      }
    }
    __syncthreads();  // next tile's first K-tile staged; lists settled
    cur ^= 1;
  }

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
// 256x256-tile counted-pipeline variant (guide T3+T4 adapted).
//
// STATUS: EXPERIMENTAL, probe-only (tools/knn_probe.hip modes 4/5); NOT
// wired into the torch binding. Measured A/B (B=4096 x N=2M x D=768):
// gemm-only 809 TF vs the 128^2 kernel's 853 TF — this adaptation of the
// guide's 8-phase template (2-phase windows at BK=32, 3 slots, counted
// vmcnt(4)) does not reproduce the template's pipelining gains; the
// finer per-phase ds_read/MFMA/glds interleave appears essential.
// Kept for the next optimization round.
//
// Geometry: block = 512 threads (8 waves as 2 row-halves x 4 col-quads),
// output tile 256 queries x 256 corpus rows; per-wave output 128x64
// (acc 8x4 fragments). K advances in 32-deep windows ("K-tiles"), each
// split into 2 phases (one n-half x all m x 16 MFMA). Staging: 3 LDS
// slots of (A 16 KiB + B 16 KiB); each wave stages 2 KiB of A and 2 KiB
// of B per window, issued TWO windows ahead of use, so the per-window
// `s_waitcnt vmcnt(4)` certifies the incoming K-tile while the next one
// stays in flight across the RAW barriers (never a vmcnt(0) drain in the
// main loop). Certification is cross-wave safe because every wave waits
// vmcnt(4) at the window's first phase and all reads happen after that
// phase's barrier.
//
// A/B images per K-tile: [256 rows][32 k] bf16, 64-byte rows; fragment
// reads are ds_read_b128 with the 16-B slot swizzled by ((row>>2)&3) on
// both the glds source and the read (rule 21).
// ===========================================================================

constexpr int BM2 = 256;
constexpr int BN2 = 256;
constexpr int BK2 = 32;
constexpr int THREADS2 = 512;
constexpr int IMG_BYTES2 = BM2 * BK2 * 2;       // 16 KiB per operand image
constexpr int SLOT_BYTES2 = 2 * IMG_BYTES2;     // A+B per K-tile
constexpr int NSLOT2 = 3;

DEVINL void stage_piece2(const bf16_t* __restrict__ src, int row0, int row_max,
                         long row_bytes, int ktile_byte, char* img_base,
                         int piece_off, int lane) {
  // one 1 KiB piece: wave-uniform LDS base + lane*16; 64-B image rows
  const int P = piece_off + lane * 16;
  const int r = P >> 6;
  const int s_phys = (P >> 4) & 3;
  const int s_log = s_phys ^ ((r >> 2) & 3);
  const int gr = min(row0 + r, row_max);
  const char* gaddr =
      (const char*)src + (size_t)gr * row_bytes + ktile_byte + s_log * 16;
  glds16(gaddr, img_base + piece_off);
}

DEVINL bf16x8 read_frag2(const char* img, int row, int slot) {
  const int s_phys = slot ^ ((row >> 2) & 3);
  return *(const bf16x8*)(img + row * 64 + s_phys * 16);
}

template <int EPI_MODE>  // 0 = full, 1 = GEMM only
__global__ __launch_bounds__(THREADS2, 2) void cosine_topk_partial256_t(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ C,
    float* __restrict__ partial_score, int* __restrict__ partial_idx,
    int B, int N, int D, int chunk_tiles, int nchunks,
    unsigned* rowthr = nullptr, unsigned long long* stats = nullptr) {
  __shared__ char smem[NSLOT2 * SLOT_BYTES2 + 2 * BM2 * KMAX * 4];
  char* const smem0 = smem;
  auto aimg = [&](int slot) -> char* { return smem0 + slot * SLOT_BYTES2; };
  auto bimg = [&](int slot) -> char* {
    return smem0 + slot * SLOT_BYTES2 + IMG_BYTES2;
  };
  float* lsc = (float*)(smem + NSLOT2 * SLOT_BYTES2);
  int* lix = (int*)(smem + NSLOT2 * SLOT_BYTES2 + BM2 * KMAX * 4);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int g = lane >> 4;
  const int cl = lane & 15;
  const int wr = wid >> 2;  // row half (0/1): rows wr*128..+128
  const int wc = wid & 3;   // col quad (0..3): cols wc*64..+64

  // XCD-aware remap (see the 128^2 kernel)
  const int nrt = gridDim.y;
  int chunk_id, row_tile;
  if ((gridDim.x & 7) == 0 && gridDim.x * nrt >= 512) {
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

  const int row0 = row_tile * BM2;
  const long rb = (long)D * 2;
  const int ntiles_total = (N + BN2 - 1) / BN2;
  const int tile0 = chunk_id * chunk_tiles;
  const int tiles_here = min(chunk_tiles, ntiles_total - tile0);
  const int nkt = D / BK2;  // windows per col tile (24 at D=768)

  for (int i = tid; i < BM2 * KMAX; i += THREADS2) {
    lsc[i] = NEG_INF;
    lix[i] = -1;
  }
  __syncthreads();

  // stage one window's worth for (global K-tile kt of col tile jt)
  auto stage_window = [&](int jt, int kt) {
    if (jt >= tiles_here) return;
    const int slot = (jt * nkt + kt) % NSLOT2;
    const int kb = kt * BK2 * 2;
    const int col0 = (tile0 + jt) * BN2;
    // A: half(wr) rows wr*128..+128; wave's 2 KiB at + wc*2048
    const int aoff = wr * 8192 + wc * 2048;
    stage_piece2(Q, row0, B - 1, rb, kb, aimg(slot), aoff, lane);
    stage_piece2(Q, row0, B - 1, rb, kb, aimg(slot), aoff + 1024, lane);
    // B: quad(wc) cols wc*64..+64; wave's 2 KiB at + wr*2048
    const int boff = wc * 4096 + wr * 2048;
    stage_piece2(C, col0, N - 1, rb, kb, bimg(slot), boff, lane);
    stage_piece2(C, col0, N - 1, rb, kb, bimg(slot), boff + 1024, lane);
  };

  // prologue: first two windows in flight
  stage_window(0, 0);
  stage_window(0, min(1, nkt - 1));
  // NB: nkt >= 2 always (D >= 64)

  for (int j = 0; j < tiles_here; ++j) {
    const int col0 = (tile0 + j) * BN2;

    float warm = NEG_INF;  // per-lane warm threshold for rows wr*128+{lane, 64+lane}
    float warm2 = NEG_INF;
    if (EPI_MODE != 1 && rowthr != nullptr) {
      const int r1 = row0 + wr * 128 + lane;
      const int r2 = r1 + 64;
      if (r1 < B) warm = dec_f32(rowthr[r1]);
      if (r2 < B) warm2 = dec_f32(rowthr[r2]);
    }

    f32x4 acc[8][4];
#pragma unroll
    for (int m = 0; m < 8; ++m)
#pragma unroll
      for (int n = 0; n < 4; ++n) acc[m][n] = f32x4{0.f, 0.f, 0.f, 0.f};

    for (int kt = 0; kt < nkt; ++kt) {
      const int slot = (j * nkt + kt) % NSLOT2;
      const char* Ai = aimg(slot) + wr * 8192;          // wave's A half
      const char* Bi = bimg(slot) + wc * 4096;          // wave's B quad
#pragma unroll
      for (int ph = 0; ph < 2; ++ph) {
        if (ph == 0) {
          // certify K-tile kt (its 4 pieces are the oldest outstanding);
          // K-tile kt+1's 4 glds stay in flight across the raw barrier
          asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        }
        {
          // stage half of the (kt+2) window per phase
          const int kt2 = kt + 2;
          const int jt2 = j + kt2 / nkt;
          const int kk2 = kt2 % nkt;
          if (ph == 0) {
            if (jt2 < tiles_here) {
              const int s2 = (jt2 * nkt + kk2) % NSLOT2;
              const int kb2 = kk2 * BK2 * 2;
              const int c2 = (tile0 + jt2) * BN2;
              const int aoff = wr * 8192 + wc * 2048;
              stage_piece2(Q, row0, B - 1, rb, kb2, aimg(s2), aoff, lane);
              stage_piece2(Q, row0, B - 1, rb, kb2, aimg(s2), aoff + 1024, lane);
              (void)c2;
            }
          } else {
            if (jt2 < tiles_here) {
              const int s2 = (jt2 * nkt + kk2) % NSLOT2;
              const int kb2 = kk2 * BK2 * 2;
              const int c2 = (tile0 + jt2) * BN2;
              const int boff = wc * 4096 + wr * 2048;
              stage_piece2(C, c2, N - 1, rb, kb2, bimg(s2), boff, lane);
              stage_piece2(C, c2, N - 1, rb, kb2, bimg(s2), boff + 1024, lane);
            }
          }
        }
        __builtin_amdgcn_s_barrier();  // pieces certified for every wave
        bf16x8 af[8], bf[2];
#pragma unroll
        for (int m = 0; m < 8; ++m)
          af[m] = read_frag2(Ai, m * 16 + cl, g);
#pragma unroll
        for (int n = 0; n < 2; ++n)
          bf[n] = read_frag2(Bi, (ph * 2 + n) * 16 + cl, g);
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int m = 0; m < 8; ++m)
#pragma unroll
          for (int n = 0; n < 2; ++n)
            acc[m][ph * 2 + n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[m], bf[n], acc[m][ph * 2 + n], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
        __builtin_amdgcn_s_barrier();  // LDS slot reuse guard
      }
    }

    // ---- top-k epilogue (same design as the 128^2 kernel) ---------------
    if constexpr (EPI_MODE == 1) {
#pragma unroll
      for (int m = 0; m < 8; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          asm volatile("" ::"v"(acc[m][n]));
    } else {
      const int colb = col0 + wc * 64 + cl;
      // lists are shared across the 4 col-quads of a row: serialise by wc
#pragma unroll
      for (int phw = 0; phw < 4; ++phw) {
        if (wc == phw) {
#pragma unroll
          for (int m = 0; m < 8; ++m) {
#pragma unroll
            for (int reg = 0; reg < 4; ++reg) {
              const int row = wr * 128 + m * 16 + g * 4 + reg;
              const int rl = m * 16 + g * 4 + reg;  // 0..127 within half
              const float rwarm =
                  __shfl(rl < 64 ? warm : warm2, rl & 63, 64);
              const int lbase = row * KMAX;
              const float rmin0 = fmaxf(lsc[lbase], rwarm);
              float w0 = (colb + 0 < N) ? acc[m][0][reg] : NEG_INF;
              float w1 = (colb + 16 < N) ? acc[m][1][reg] : NEG_INF;
              float w2 = (colb + 32 < N) ? acc[m][2][reg] : NEG_INF;
              float w3 = (colb + 48 < N) ? acc[m][3][reg] : NEG_INF;
              float gmax = fmaxf(fmaxf(w0, w1), fmaxf(w2, w3));
#pragma unroll
              for (int off = 1; off < 16; off <<= 1)
                gmax = fmaxf(gmax, __shfl_xor(gmax, off, 64));
              if (gmax > rmin0) {
                topk_extract_group<true>(lsc, lix, lbase, rwarm, w0, w1, w2,
                                         w3, colb, N, lane, g, rowthr,
                                         (row0 + row < B) ? row0 + row + 1 : 0);
              }
            }
          }
        }
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();  // raw: cross-tile glds stay in flight
      }
    }
  }

  // write partials: [B][nchunks][KMAX]
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  if (tid < BM2) {
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
  if (tid >= BM2 && tid < 2 * BM2) {
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

}  // namespace kakveda

