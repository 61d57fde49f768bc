// Weight-gradient (wrw) v2 for 5x5 stride-1 NHWC conv, CI=16, on gfx950.
//
// dW[kh,kw,ci][o] = sum_{n,ho,wo} in[n,ho+kh,wo+kw,ci] * gout[n,ho,wo,o]
//
// v1 (convwrw.hip) stages TRANSPOSED images with per-element ds writes
// and gathers the shifted A fragment with an alignbyte funnel; it is
// LDS-gather bound (1.10 ms standalone on the conv2 shape vs MIOpen's
// 1.60 in-step). v2 removes both costs:
//
//   * LDS images keep the NATURAL pixel-major layout ([pix][ci] for the
//     input — identical to global NHWC — and per-o-tile [pix][16] for
//     gout), so staging is pure LDS-DMA (__builtin_amdgcn_global_load_lds,
//     16 B/lane): zero VALU work, zero ds_write instructions. Lanes whose
//     pixel falls in the zero-padding read a device zero page, so padding
//     is zero-filled in the same pass.
//   * MFMA fragments come from ds_read_b64_tr_b16 (gfx950's LDS
//     transpose read, cdna_hip_programming.md T10): each 16-lane group
//     reads one 4(pix)x16(chan) row-major tile — lane g supplies
//     tile_base + g*8 (8B-aligned) and receives CHANNEL COLUMN g across
//     the 4 pixels. Both A and B fragments use the SAME k-slot->pixel
//     permutation (slot q*8+j <-> pixel 4q+j / 16+4q+j), so the MFMA
//     contraction pairs matched pixels. The kw shift of A is a whole-
//     pixel base offset — no misaligned gathers, no funnel shifts.
//
// Work layout: grid-persistent workgroups own row-blocks of WRW2_R gout
// rows; 4 waves split the 25 (kh,kw) tap tiles round-robin and each
// wave computes BOTH o-tiles for its taps (A-fragment reuse), keeping
// all accumulators in VGPRs until one final per-workgroup slab flush
// (partials summed in torch, same contract as v1).

#include <hip/hip_runtime.h>

#include <cstdint>

typedef unsigned short bf16_t;
typedef __attribute__((ext_vector_type(4))) short bf16x4;
typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define WRW2_THREADS 256  // 4 waves
#define WRW2_R 2          // gout rows per block

__device__ __attribute__((aligned(64))) bf16_t g_wrw2_zeros[1024];

// ds_read_b64_tr_b16: the 16-lane group covers one 4x16 bf16 row-major
// tile (lane g supplies tile_base + g*8); lane g receives column g
// (4 bf16, one per tile row). `a` is this lane's LDS byte address.
__device__ __forceinline__ bf16x4 tr16_b64(unsigned a) {
  bf16x4 d;
  asm volatile("ds_read_b64_tr_b16 %0, %1"
               : "=v"(d)
               : "v"(a)
               : "memory");
  return d;
}

__device__ __forceinline__ unsigned lds_addr(const void* p) {
  return (unsigned)(uintptr_t)(const __attribute__((address_space(3)))
                               void*)p;
}

template <int COT, int PIX>
__global__ __launch_bounds__(WRW2_THREADS) void k_conv5_wrw16_nhwc(
    const bf16_t* __restrict__ in,    // [N][Hi][Wi][16]
    const bf16_t* __restrict__ gout,  // [N][Ho][Wo][CO]
    float* __restrict__ part,         // [nWG][T16][CO]
    int Nn, int Hi, int Wi, int Ho, int Wo) {
  constexpr int CI = 16;
  constexpr int CO = COT * 16;
  constexpr int NT = 25;                 // tap tiles = (kh,kw) pairs
  // +1 slab tile: the spare (wid 1, j 6) slot computes the BIAS grad
  // with an all-ones A row (D[0][o] = sum_k B[k][o] accumulated over
  // every pixel window) — kills the separate at::native bias reduce
  constexpr int T16 = (NT + 1) * 16;
  // row region: one 16-wide subimage row, padded to whole 1 KiB LDS-DMA
  // chunks so every glds writes full 64-lane spans (no partial EXEC)
  constexpr int ROW_BYTES = PIX * CI * 2;
  constexpr int NCH = (ROW_BYTES + 1023) / 1024;       // glds chunks/row
  constexpr int RPB = NCH * 1024;                      // region bytes
  constexpr int NROW_IN = WRW2_R + 4;
  constexpr int NROW_GO = WRW2_R * COT;
  __shared__ __attribute__((aligned(128))) char lds_all[(NROW_IN + NROW_GO) *
                                                        RPB];

  const int lane = threadIdx.x & 63;
  const int q = lane >> 4;
  const int m = lane & 15;
  const int wid = threadIdx.x >> 6;

  const int W = (Wo + 31) >> 5;          // 32-pixel K windows per row
  const int blocks_h = (Ho + WRW2_R - 1) / WRW2_R;
  const long long n_blocks = (long long)Nn * blocks_h;
  const int wg = blockIdx.x;
  const int n_wg = gridDim.x;

  const unsigned lds0 = lds_addr(lds_all);

  // wave-owned tap tiles: tt = wid + 4*j, accumulators for BOTH o-tiles
  constexpr int MAXT = (NT + 3) / 4;     // 7
  f32x4 acc[MAXT][COT];
#pragma unroll
  for (int j = 0; j < MAXT; ++j)
#pragma unroll
    for (int t = 0; t < COT; ++t) acc[j][t] = (f32x4)0.0f;

  for (long long blk = wg; blk < n_blocks; blk += n_wg) {
    const int bh = (int)(blk % blocks_h);
    const long long n = blk / blocks_h;
    const int ho0 = bh * WRW2_R;
    const int nrows = (Ho - ho0) < WRW2_R ? (Ho - ho0) : WRW2_R;
    const int irows = nrows + 4;

    // ---- stage via LDS-DMA: chunks round-robin over the 4 waves.
    // A rows: a global NHWC row IS the LDS [pix][16] row (straight copy).
    {
      const int nchunks_a = irows * NCH;
      for (int t = wid; t < nchunks_a; t += 4) {
        const int ir = t / NCH;
        const int c = t - ir * NCH;
        const int slot = c * 64 + lane;              // 16B granule index
        const int pix = slot >> 1;
        const bf16_t* src =
            (pix < Wi) ? in + ((n * Hi + (ho0 + ir)) * (long long)Wi * CI +
                               (long long)slot * 8)
                       : g_wrw2_zeros;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)src,
            (__attribute__((address_space(3))) void*)(lds_all + ir * RPB +
                                                      c * 1024),
            16, 0, 0);
      }
      // gout rows: global [pix][CO] -> per-o-tile LDS [pix][16]
      const int nchunks_b = nrows * COT * NCH;
      for (int t = wid; t < nchunks_b; t += 4) {
        const int rr = t / (COT * NCH);
        const int rem = t - rr * (COT * NCH);
        const int ot = rem / NCH;
        const int c = rem - ot * NCH;
        const int slot = c * 64 + lane;
        const int pix = slot >> 1;
        const int half = slot & 1;
        const bf16_t* src =
            (pix < Wo)
                ? gout + ((n * Ho + (ho0 + rr)) * (long long)Wo * CO +
                          (long long)pix * CO + ot * 16 + half * 8)
                : g_wrw2_zeros;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)src,
            (__attribute__((address_space(3))) void*)(lds_all +
                                                      (NROW_IN + rr * COT +
                                                       ot) * RPB +
                                                      c * 1024),
            16, 0, 0);
      }
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();

    for (int r = 0; r < nrows; ++r) {
      // A-row base addresses for this wave's 7 tap tiles (tt clamped so
      // the unused 7th tile of waves 1-3 still reads in-bounds LDS; its
      // MFMA is skipped)
      unsigned abase[MAXT];
#pragma unroll
      for (int j = 0; j < MAXT; ++j) {
        int tt = wid + 4 * j;
        if (tt >= NT) tt = NT - 1;
        const int kh = tt / 5;
        const int kw = tt - kh * 5;
        abase[j] = lds0 + (r + kh) * RPB + (kw + 4 * q) * 32 + m * 8;
      }
      const unsigned bbase0 = lds0 + (NROW_IN + r * COT + 0) * RPB +
                              (4 * q) * 32 + m * 8;
      const unsigned bbase1 = lds0 + (NROW_IN + r * COT + (COT - 1)) * RPB +
                              (4 * q) * 32 + m * 8;
      for (int w = 0; w < W; ++w) {
        const unsigned poff = (unsigned)(w * 32 * 32);  // pix0 bytes
        // ONE asm block issues every tr read for this (r,w) and ends
        // with the lgkmcnt drain: the compiler materialises the
        // bf16x4->bf16x8 repack VALU at an asm OUTPUT's def site, so
        // any multi-asm structure lets it consume un-landed tr
        // destinations BEFORE a separate wait (observed: v_lshrrev/
        // v_perm ahead of s_waitcnt -> corrupt fragments). Outputs are
        // early-clobber so no read's destination aliases a later-used
        // address register.
        bf16x4 bfr[2][2];
        bf16x4 afr[MAXT][2];
        asm volatile(
            "ds_read_b64_tr_b16 %0, %18\n\t"
            "ds_read_b64_tr_b16 %1, %18 offset:512\n\t"
            "ds_read_b64_tr_b16 %2, %19\n\t"
            "ds_read_b64_tr_b16 %3, %19 offset:512\n\t"
            "ds_read_b64_tr_b16 %4, %20\n\t"
            "ds_read_b64_tr_b16 %5, %20 offset:512\n\t"
            "ds_read_b64_tr_b16 %6, %21\n\t"
            "ds_read_b64_tr_b16 %7, %21 offset:512\n\t"
            "ds_read_b64_tr_b16 %8, %22\n\t"
            "ds_read_b64_tr_b16 %9, %22 offset:512\n\t"
            "ds_read_b64_tr_b16 %10, %23\n\t"
            "ds_read_b64_tr_b16 %11, %23 offset:512\n\t"
            "ds_read_b64_tr_b16 %12, %24\n\t"
            "ds_read_b64_tr_b16 %13, %24 offset:512\n\t"
            "ds_read_b64_tr_b16 %14, %25\n\t"
            "ds_read_b64_tr_b16 %15, %25 offset:512\n\t"
            "ds_read_b64_tr_b16 %16, %26\n\t"
            "ds_read_b64_tr_b16 %17, %26 offset:512\n\t"
            "s_waitcnt lgkmcnt(0)"
            : "=&v"(bfr[0][0]), "=&v"(bfr[0][1]), "=&v"(bfr[1][0]),
              "=&v"(bfr[1][1]), "=&v"(afr[0][0]), "=&v"(afr[0][1]),
              "=&v"(afr[1][0]), "=&v"(afr[1][1]), "=&v"(afr[2][0]),
              "=&v"(afr[2][1]), "=&v"(afr[3][0]), "=&v"(afr[3][1]),
              "=&v"(afr[4][0]), "=&v"(afr[4][1]), "=&v"(afr[5][0]),
              "=&v"(afr[5][1]), "=&v"(afr[6][0]), "=&v"(afr[6][1])
            : "v"(bbase0 + poff), "v"(bbase1 + poff),
              "v"(abase[0] + poff), "v"(abase[1] + poff),
              "v"(abase[2] + poff), "v"(abase[3] + poff),
              "v"(abase[4] + poff), "v"(abase[5] + poff),
              "v"(abase[6] + poff)
            : "memory");
        __builtin_amdgcn_sched_barrier(0);
#pragma unroll
        for (int j = 0; j < MAXT; ++j) {
          const int tt = wid + 4 * j;
          const bool bias_tile = (tt == NT);   // spare slot: wid 1, j 6
          if (tt >= NT && !bias_tile) continue;
          union { struct { bf16x4 lo, hi; } p; bf16x8 v; } a;
          if (bias_tile) {
            const bf16x8 ones = (bf16x8)(short)0x3F80;  // bf16 1.0 splat
            a.v = (m == 0) ? ones : (bf16x8)(short)0;
          } else {
            a.p.lo = afr[j][0];
            a.p.hi = afr[j][1];
          }
#pragma unroll
          for (int ot = 0; ot < COT; ++ot) {
            union { struct { bf16x4 lo, hi; } p; bf16x8 v; } b;
            b.p.lo = bfr[ot][0];
            b.p.hi = bfr[ot][1];
            acc[j][ot] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                a.v, b.v, acc[j][ot], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();  // before the next block restages LDS
  }

  // ---- flush accumulators: part[wg][tap][o]
  // C/D layout of mfma_f32_16x16x32_bf16: lane (q,m), reg i ->
  // row = q*4 + i (the A row = tap-within-tile), col = m (the B col = o)
  float* base = part + (long long)wg * T16 * CO;
#pragma unroll
  for (int j = 0; j < MAXT; ++j) {
    const int tt = wid + 4 * j;
    if (tt > NT) continue;               // NT itself = the bias tile
#pragma unroll
    for (int ot = 0; ot < COT; ++ot) {
      const int o = ot * 16 + m;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int tap = tt * 16 + q * 4 + i;
        base[(long long)tap * CO + o] = acc[j][ot][i];
      }
    }
  }
}

// ---------------------------------------------------------------------
// probe: verify the tr16 lane mapping numerically (test_kernels_gpu).
// in : 64 bf16 = one 4x16 row-major tile. out[lane][j] = element j of
// lane `lane`'s tr read — expected out[l][j] == tile[j][l&15] for the
// first 16-lane group (all groups read the same tile here).
// ---------------------------------------------------------------------
__global__ void k_tr16_probe(const bf16_t* __restrict__ in,
                             bf16_t* __restrict__ out) {
  __shared__ __attribute__((aligned(128))) bf16_t tile[64];
  const int lane = threadIdx.x & 63;
  tile[lane] = in[lane];
  __syncthreads();
  const unsigned a = lds_addr(tile) + (lane & 15) * 8;
  bf16x4 d;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=&v"(d) : "v"(a) : "memory");
#pragma unroll
  for (int j = 0; j < 4; ++j) out[lane * 4 + j] = d[j];
}

// ---------------------------------------------------------------------
// CI=4 variant (conv1: 4 padded channels -> 16 outputs).
//
// The CI=16 scheme needs 16-wide image rows so that a tr-read's j step
// is ONE k-pixel on both operands. With 4 channels the natural image
// rows are 4-wide, so instead we stage a KH-INTERLEAVED image
//     I_p[pix][kho][ci] = in[row p+kho][pix][ci]   (kho = 0..3)
// whose 16-wide rows make lane m = (kho*4+ci) the channel column and
// keep the 1-pixel j step. Tap tiles become t = kw*2 + khg
// (kw = 0..4, khg = 0..1, kh = khg*4+kho): 10 tiles of 16 slots hold
// the 100 real taps (kh>4 slots are dead weight, 38%; their values are
// whatever the clamped staging rows hold and are dropped at unpack).
// Four interleaved images (p = r + 4*khg for r in 0..1, khg in 0..1)
// are filled by 4-byte LDS-DMA granules (dest byte 4l maps to
// pix = l>>3, kho = (l>>1)&3, ci-pair = l&1 — per-lane global source
// does the interleave, dest stays lane-linear).
// ---------------------------------------------------------------------

template <int PIX>
__global__ __launch_bounds__(WRW2_THREADS) void k_conv5_wrw4_nhwc(
    const bf16_t* __restrict__ in,    // [N][Hi][Wi][4]
    const bf16_t* __restrict__ gout,  // [N][Ho][Wo][16]
    float* __restrict__ part,         // [nWG][176][16]
    int Nn, int Hi, int Wi, int Ho, int Wo) {
  // v3: the transpose read gathers PER-LANE 8B quarters (verified by
  // scripts/tr16_scatter_probe.py: out[m][j] = mem[addr[4j+(m>>2)] +
  // (m&3) elems]), so the A fragment (lane m = kho*4+ci) reads STRAIGHT
  // from the 6 raw [pix][4] input rows: lane l points at
  // row(r + khg*4 + (l&3)), pixel (pix0 + kw + 4q + (l>>2)&3).
  // No kh-interleaved images, no 4x staging duplication: per block the
  // A staging is 6 rows x PIX*8B once. Row regions are skewed by 64 B
  // so the 4 rows a group touches land on distinct LDS banks.
  constexpr int CI = 4;
  constexpr int CO = 16;
  constexpr int NT = 10;                  // (kw 0..4) x (khg 0..1)
  constexpr int T16 = (NT + 1) * 16;      // +bias tile (wid 2, j 2)
  constexpr int AROW_BYTES = PIX * CI * 2;
  constexpr int NCHA = (AROW_BYTES + 1023) / 1024;
  constexpr int ARPB = NCHA * 1024 + 64;  // 64B skew: distinct banks/row
  constexpr int BROW_BYTES = PIX * 16 * 2;
  constexpr int NCHB = (BROW_BYTES + 1023) / 1024;
  constexpr int BRPB = NCHB * 1024;
  constexpr int NROW_A = 6;               // rows ho0 .. ho0+5
  __shared__ __attribute__((aligned(128))) char lds_all[NROW_A * ARPB +
                                                        WRW2_R * BRPB];
  constexpr int BOFF = NROW_A * ARPB;

  const int lane = threadIdx.x & 63;
  const int q = lane >> 4;
  const int m = lane & 15;
  const int wid = threadIdx.x >> 6;

  const int W = (Wo + 31) >> 5;
  const int blocks_h = (Ho + WRW2_R - 1) / WRW2_R;
  const long long n_blocks = (long long)Nn * blocks_h;
  const int wg = blockIdx.x;
  const int n_wg = gridDim.x;
  const unsigned lds0 = lds_addr(lds_all);

  constexpr int MAXT = 3;
  f32x4 acc[MAXT];
#pragma unroll
  for (int j = 0; j < MAXT; ++j) acc[j] = (f32x4)0.0f;

  for (long long blk = wg; blk < n_blocks; blk += n_wg) {
    const int bh = (int)(blk % blocks_h);
    const long long n = blk / blocks_h;
    const int ho0 = bh * WRW2_R;
    const int nrows = (Ho - ho0) < WRW2_R ? (Ho - ho0) : WRW2_R;

    // ---- stage 6 raw input rows + gout rows (16B LDS-DMA granules)
    {
      const int nchunks_a = NROW_A * NCHA;
      for (int t = wid; t < nchunks_a; t += 4) {
        const int ir = t / NCHA;
        const int c = t - ir * NCHA;
        const int slot = c * 64 + lane;              // 16B = 2 pixels
        const int pix = slot * 2;
        int row = ho0 + ir;
        if (row >= Hi) row = Hi - 1;                 // tail blocks only
        const bf16_t* src =
            (pix + 1 < Wi) ? in + ((n * Hi + row) * (long long)Wi * CI +
                                   (long long)pix * CI)
                           : g_wrw2_zeros;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)src,
            (__attribute__((address_space(3))) void*)(lds_all + ir * ARPB +
                                                      c * 1024),
            16, 0, 0);
      }
      const int nchunks_b = nrows * NCHB;
      for (int t = wid; t < nchunks_b; t += 4) {
        const int rr = t / NCHB;
        const int c = t - rr * NCHB;
        const int slot = c * 64 + lane;
        const int pix = slot >> 1;
        const bf16_t* src =
            (pix < Wo) ? gout + ((n * Ho + (ho0 + rr)) * (long long)Wo * CO +
                                 (long long)slot * 8)
                       : g_wrw2_zeros;
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) void*)src,
            (__attribute__((address_space(3))) void*)(lds_all + BOFF +
                                                      rr * BRPB + c * 1024),
            16, 0, 0);
      }
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();

    for (int r = 0; r < nrows; ++r) {
      // per-lane A bases: lane quarter l&15 supplies
      // row(r + khg*4 + (lane&3)), pixel (kw + 4q + ((lane&15)>>2))
      unsigned abase[MAXT];
#pragma unroll
      for (int j = 0; j < MAXT; ++j) {
        int t = wid + 4 * j;
        if (t >= NT) t = NT - 1;
        const int kw = t >> 1;
        const int khg = t & 1;
        int rowsel = r + khg * 4 + (lane & 3);
        if (rowsel > NROW_A - 1) rowsel = NROW_A - 1;  // dead taps only
        abase[j] = lds0 + rowsel * ARPB +
                   (unsigned)((kw + 4 * q + ((lane & 15) >> 2)) * 8);
      }
      const unsigned bbase = lds0 + BOFF + r * BRPB + (4 * q) * 32 + m * 8;
      for (int w = 0; w < W; ++w) {
        const unsigned poffB = (unsigned)(w * 32 * 32);
        const unsigned poffA = (unsigned)(w * 32 * 8);
        bf16x4 fr[8];
        asm volatile(
            "ds_read_b64_tr_b16 %0, %8\n\t"
            "ds_read_b64_tr_b16 %1, %8 offset:512\n\t"
            "ds_read_b64_tr_b16 %2, %9\n\t"
            "ds_read_b64_tr_b16 %3, %9 offset:128\n\t"
            "ds_read_b64_tr_b16 %4, %10\n\t"
            "ds_read_b64_tr_b16 %5, %10 offset:128\n\t"
            "ds_read_b64_tr_b16 %6, %11\n\t"
            "ds_read_b64_tr_b16 %7, %11 offset:128\n\t"
            "s_waitcnt lgkmcnt(0)"
            : "=&v"(fr[0]), "=&v"(fr[1]), "=&v"(fr[2]), "=&v"(fr[3]),
              "=&v"(fr[4]), "=&v"(fr[5]), "=&v"(fr[6]), "=&v"(fr[7])
            : "v"(bbase + poffB), "v"(abase[0] + poffA),
              "v"(abase[1] + poffA), "v"(abase[2] + poffA)
            : "memory");
        __builtin_amdgcn_sched_barrier(0);
        union { struct { bf16x4 lo, hi; } p; bf16x8 v; } b;
        b.p.lo = fr[0];
        b.p.hi = fr[1];
#pragma unroll
        for (int j = 0; j < MAXT; ++j) {
          const int t = wid + 4 * j;
          const bool bias_tile = (t == NT);  // spare slot: wid 2, j 2
          if (t >= NT && !bias_tile) continue;
          union { struct { bf16x4 lo, hi; } p; bf16x8 v; } a;
          if (bias_tile) {
            const bf16x8 ones = (bf16x8)(short)0x3F80;
            a.v = (m == 0) ? ones : (bf16x8)(short)0;
          } else {
            a.p.lo = fr[2 + 2 * j];
            a.p.hi = fr[3 + 2 * j];
          }
          acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a.v, b.v,
                                                           acc[j], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // ---- flush: part[wg][t*16 + m][o], D row = q*4+i = slot m, col = o
  float* base = part + (long long)wg * T16 * CO;
#pragma unroll
  for (int j = 0; j < MAXT; ++j) {
    const int t = wid + 4 * j;
    if (t > NT) continue;                // NT itself = the bias tile
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int slot = t * 16 + q * 4 + i;
      base[(long long)slot * CO + m] = acc[j][i];
    }
  }
}

// debug: run block 0's staging for (n=0, ho0=0) and dump the raw LDS
// bytes so tests can check the glds images independently of the MFMAs.
template <int COT, int PIX>
__global__ void k_wrw2_dump(const bf16_t* __restrict__ in,
                            const bf16_t* __restrict__ gout,
                            bf16_t* __restrict__ dump,
                            int Nn, int Hi, int Wi, int Ho, int Wo) {
  constexpr int CI = 16;
  constexpr int CO = COT * 16;
  constexpr int ROW_BYTES = PIX * CI * 2;
  constexpr int NCH = (ROW_BYTES + 1023) / 1024;
  constexpr int RPB = NCH * 1024;
  constexpr int NROW_IN = WRW2_R + 4;
  constexpr int NROW_GO = WRW2_R * COT;
  __shared__ __attribute__((aligned(128))) char lds_all[(NROW_IN + NROW_GO) *
                                                        RPB];
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int nrows = Ho < WRW2_R ? Ho : WRW2_R;
  const int irows = nrows + 4;
  {
    const int nchunks_a = irows * NCH;
    for (int t = wid; t < nchunks_a; t += 4) {
      const int ir = t / NCH;
      const int c = t - ir * NCH;
      const int slot = c * 64 + lane;
      const int pix = slot >> 1;
      const bf16_t* src = (pix < Wi)
          ? in + ((long long)ir * Wi * CI + (long long)slot * 8)
          : g_wrw2_zeros;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(lds_all + ir * RPB +
                                                    c * 1024),
          16, 0, 0);
    }
    const int nchunks_b = nrows * COT * NCH;
    for (int t = wid; t < nchunks_b; t += 4) {
      const int rr = t / (COT * NCH);
      const int rem = t - rr * (COT * NCH);
      const int ot = rem / NCH;
      const int c = rem - ot * NCH;
      const int slot = c * 64 + lane;
      const int pix = slot >> 1;
      const int half = slot & 1;
      const bf16_t* src = (pix < Wo)
          ? gout + ((long long)rr * Wo * CO + (long long)pix * CO +
                    ot * 16 + half * 8)
          : g_wrw2_zeros;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(lds_all +
                                                    (NROW_IN + rr * COT +
                                                     ot) * RPB + c * 1024),
          16, 0, 0);
    }
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  const bf16_t* l16 = (const bf16_t*)lds_all;
  for (int i = threadIdx.x; i < (NROW_IN + NROW_GO) * RPB / 2;
       i += blockDim.x)
    dump[i] = l16[i];
}


// probe 2: SCATTERED per-lane addresses. If the transpose read gathers
// each lane's own 8B quarter (out[g][j] = mem[addr[4j+(g>>2)] + 2*(g&3)])
// the wrw4 staging can drop its interleaved-image duplication. in: 512
// bf16 of content; addr_off[l] (elements) programs lane l's address.
__global__ void k_tr16_probe2(const bf16_t* __restrict__ in,
                              const int* __restrict__ addr_off,
                              bf16_t* __restrict__ out) {
  __shared__ __attribute__((aligned(128))) bf16_t buf[512];
  const int lane = threadIdx.x & 63;
  for (int i = threadIdx.x; i < 512; i += blockDim.x) buf[i] = in[i];
  __syncthreads();
  const unsigned a = lds_addr(buf) + (unsigned)addr_off[lane] * 2;
  bf16x4 d;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=&v"(d) : "v"(a) : "memory");
#pragma unroll
  for (int j = 0; j < 4; ++j) out[lane * 4 + j] = d[j];
}

extern "C" {

int geops_conv5_wrw16_nhwc(const bf16_t* in, const bf16_t* gout, float* part,
                           int Nn, int Hi, int Wi, int Ho, int Wo, int CI,
                           int CO, int n_wg, hipStream_t s) {
  if (CI != 16) return -1;
  const int W = (Wo + 31) >> 5;
  const int need = (W * 32 + 4) > Wi ? (W * 32 + 4) : Wi;
#define W2LAUNCH(COT_, PIX_)                                              \
  if (need <= PIX_) {                                                     \
    hipLaunchKernelGGL((k_conv5_wrw16_nhwc<COT_, PIX_>), dim3(n_wg),      \
                       dim3(WRW2_THREADS), 0, s, in, gout, part, Nn, Hi,  \
                       Wi, Ho, Wo);                                       \
    return n_wg;                                                          \
  }
  if (CO == 32) { W2LAUNCH(2, 136) W2LAUNCH(2, 232) }
  if (CO == 16) { W2LAUNCH(1, 136) W2LAUNCH(1, 232) }
#undef W2LAUNCH
  return -1;
}

int geops_conv5_wrw4_nhwc(const bf16_t* in, const bf16_t* gout, float* part,
                          int Nn, int Hi, int Wi, int Ho, int Wo, int CI,
                          int CO, int n_wg, hipStream_t s) {
  // Wi must be even: the A staging copies 2-pixel (16B) granules
  if (CI != 4 || CO != 16 || (Wi & 1)) return -1;
  const int W = (Wo + 31) >> 5;
  const int need = (W * 32 + 4) > Wi ? (W * 32 + 4) : Wi;
#define W4LAUNCH(PIX_)                                                    \
  if (need <= PIX_) {                                                     \
    hipLaunchKernelGGL((k_conv5_wrw4_nhwc<PIX_>), dim3(n_wg),             \
                       dim3(WRW2_THREADS), 0, s, in, gout, part, Nn, Hi,  \
                       Wi, Ho, Wo);                                       \
    return n_wg;                                                          \
  }
  W4LAUNCH(136) W4LAUNCH(232)
#undef W4LAUNCH
  return -1;
}

void geops_tr16_probe(const bf16_t* in, bf16_t* out, hipStream_t s) {
  hipLaunchKernelGGL(k_tr16_probe, dim3(1), dim3(64), 0, s, in, out);
}

void geops_tr16_probe2(const bf16_t* in, const int* addr_off, bf16_t* out,
                       hipStream_t s) {
  hipLaunchKernelGGL(k_tr16_probe2, dim3(1), dim3(64), 0, s, in, addr_off,
                     out);
}

// dump size in bf16 elements for the (COT,PIX) variant, or -1
int geops_wrw2_dump(const bf16_t* in, const bf16_t* gout, bf16_t* dump,
                    int Nn, int Hi, int Wi, int Ho, int Wo, int CO,
                    hipStream_t s) {
  if (CO == 32) {
    constexpr int RPB = ((136 * 16 * 2 + 1023) / 1024) * 1024;
    hipLaunchKernelGGL((k_wrw2_dump<2, 136>), dim3(1), dim3(WRW2_THREADS),
                       0, s, in, gout, dump, Nn, Hi, Wi, Ho, Wo);
    return (6 + 4) * RPB / 2;
  }
  return -1;
}

}  // extern "C"
