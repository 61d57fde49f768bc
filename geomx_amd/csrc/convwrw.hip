// Weight-gradient (wrw) kernel for 5x5 stride-1 NHWC convolutions on
// gfx950 MFMA — the single hottest op of the flagship CNN step under
// MIOpen (igemm_wrw ~1.6 ms for conv2).
//
// dW[kh,kw,ci][o] = sum_{n,ho,wo} in[n,ho+kh,wo+kw,ci] * gout[n,ho,wo,o]
//
// Formulation: D[M=tap][N=o], K = pixels. mfma_f32_16x16x32_bf16:
//   A[m=tap][k=pixel]  = shifted input — read as 8 x ds_read_u16
//     gathers from an LDS-transposed window in_t[(row,ci)][pix]
//     (the kw shift makes the run 2-byte-misaligned for b128)
//   B[k=pixel][n=o]    = gout — aligned ds_read_b128 from the
//     transposed row tile gout_t[o][pix]
//   D: lane l holds dW[tap=(l>>4)*4+i][o=l&15], accumulated in
//     REGISTERS per wave across the workgroup's whole row range and
//     flushed once to a per-workgroup partial slab (summed in torch).
//
// Work layout: each workgroup owns row-BLOCKS of R=4 consecutive gout
// rows of one image: the input window for the block is R+4=8 rows,
// staged (transposed) once per block; gout_t re-staged per row. Every
// (tap-tile x o-tile) pair is owned by one wave (fixed assignment), so
// accumulators never leave VGPRs. Tap order: t = (kh*5 + kw)*CI + ci.

#include <hip/hip_runtime.h>

#include <cstdint>

typedef unsigned short bf16_t;
typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define WRW_THREADS 256  // 4 waves
#define WRW_R 2          // gout rows per block (both rows' gout
                         // staged together: one barrier pair per block)

template <int CI, int COT, int PIX>
__global__ __launch_bounds__(WRW_THREADS) void k_conv5_wrw_nhwc(
    const bf16_t* __restrict__ in,    // [N][Hi][Wi][CI]
    const bf16_t* __restrict__ gout,  // [N][Ho][Wo][CO]
    float* __restrict__ part,         // [nWG][T16][CO] partial slabs
    int Nn, int Hi, int Wi, int Ho, int Wo) {
  constexpr int CO = COT * 16;
  constexpr int K = 25 * CI;             // real taps
  constexpr int NT = (K + 15) / 16;      // tap tiles
  constexpr int T16 = NT * 16;
  // LDS geometry (pitches in elements; x2 bytes must be 16B-aligned and
  // an odd number of 16B slots for conflict-free 16-row column reads).
  // PIX is a per-shape template: >= W*32+4 and >= Wi (launcher checks).
  constexpr int PIXP = PIX;
  constexpr int INPP = PIX;
  __shared__ __attribute__((aligned(16))) bf16_t
      lds_gout[WRW_R * CO * PIXP];
  __shared__ __attribute__((aligned(16))) bf16_t
      lds_in[(WRW_R + 4) * CI * INPP];

  const int lane = threadIdx.x & 63;
  const int q = lane >> 4;
  const int wid = threadIdx.x >> 6;

  const int W = (Wo + 31) >> 5;          // 32-pixel windows per row
  const int blocks_h = (Ho + WRW_R - 1) / WRW_R;
  const long long n_blocks = (long long)Nn * blocks_h;
  const int wg = blockIdx.x;
  const int n_wg = gridDim.x;

  // wave-owned (tap-tile, o-tile) pairs: pair = wid + 4*j
  constexpr int NPAIR = NT * COT;
  constexpr int MAXP = (NPAIR + 3) / 4;
  f32x4 acc[MAXP];
#pragma unroll
  for (int j = 0; j < MAXP; ++j) acc[j] = (f32x4)0.0f;

  for (long long blk = wg; blk < n_blocks; blk += n_wg) {
    const int bh = (int)(blk % blocks_h);
    const long long n = blk / blocks_h;
    const int ho0 = bh * WRW_R;
    const int nrows = (Ho - ho0) < WRW_R ? (Ho - ho0) : WRW_R;

    // ---- stage the input window, transposed: in_t[(ir*CI+ci)][pix]
    // zero the pad columns first (only [Wi, INPP) is ever garbage)
    for (int i = threadIdx.x; i < (WRW_R + 4) * CI * (INPP - Wi);
         i += blockDim.x) {
      const int r = i / (INPP - Wi);
      const int c = Wi + i % (INPP - Wi);
      lds_in[r * INPP + c] = (bf16_t)0;
    }
    const int irows = nrows + 4;
    // coalesced fill: each thread reads ONE uint4 (8 bf16) and scatters
    // 8 ds writes (the transpose) — 8x fewer global load instructions
    // than the scalar fill this replaces
    const int in_vec = (irows * Wi * CI) >> 3;  // Wi*CI % 8 == 0
    for (int ev = threadIdx.x; ev < in_vec; ev += blockDim.x) {
      const long long g0 = (long long)ev * 8;
      const int ir = (int)(g0 / (Wi * CI));
      const uint4 v = *reinterpret_cast<const uint4*>(
          in + (((n * Hi + (ho0 + ir)) * (long long)Wi * CI) +
                (g0 - (long long)ir * (Wi * CI))));
      const bf16_t* hv = reinterpret_cast<const bf16_t*>(&v);
      const int rem0 = (int)(g0 - (long long)ir * (Wi * CI));
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int rem = rem0 + j;
        const int pix = rem / CI;
        const int ci = rem - pix * CI;
        lds_in[(ir * CI + ci) * INPP + pix] = hv[j];
      }
    }

    // ---- stage ALL the block's gout rows, transposed + zero-padded
    for (int i = threadIdx.x; i < WRW_R * CO * (PIXP - Wo);
         i += blockDim.x) {
      const int rr = i / (CO * (PIXP - Wo));
      const int rem = i - rr * (CO * (PIXP - Wo));
      const int o = rem / (PIXP - Wo);
      const int c = Wo + rem % (PIXP - Wo);
      lds_gout[(rr * CO + o) * PIXP + c] = (bf16_t)0;
    }
    const int go_vec = (nrows * Wo * CO) >> 3;  // Wo*CO % 8 == 0
    for (int ev = threadIdx.x; ev < go_vec; ev += blockDim.x) {
      const long long g0 = (long long)ev * 8;
      const int rr = (int)(g0 / (Wo * CO));
      const uint4 v = *reinterpret_cast<const uint4*>(
          gout + ((n * Ho + (ho0 + rr)) * (long long)Wo * CO) +
          (g0 - (long long)rr * (Wo * CO)));
      const bf16_t* hv = reinterpret_cast<const bf16_t*>(&v);
      const int rem0 = (int)(g0 - (long long)rr * (Wo * CO));
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int rem = rem0 + j;
        const int pix = rem / CO;
        const int o = rem - pix * CO;
        lds_gout[(rr * CO + o) * PIXP + pix] = hv[j];
      }
    }
    __syncthreads();

    for (int r = 0; r < nrows; ++r) {

      // ---- accumulate this row into the wave-owned pairs
      // FULL unroll: acc[j] must be statically indexed or it spills to
      // scratch (224 B/lane measured with a rolled loop — rule 20)
#pragma unroll
      for (int j = 0; j < MAXP; ++j) {
        const int pair = wid + 4 * j;
        if (pair >= NPAIR) continue;
        const int tt = pair / COT;   // tap tile
        const int ot = pair - tt * COT;
        // this lane's tap (A row) within the tile
        const int tap = tt * 16 + (lane & 15);
        const int valid = tap < K;
        const int khkw = valid ? tap / CI : 0;
        const int ci = valid ? tap - khkw * CI : 0;
        const int kh = khkw / 5;
        const int kw = khkw - kh * 5;
        const bf16_t* arow = lds_in + ((r + kh) * CI + ci) * INPP + kw;
        const bf16_t* brow =
            lds_gout + ((r * CO) + (ot * 16) + (lane & 15)) * PIXP;
        // A-fragment base WITHOUT the kw shift: for CI >= 16 a tap tile
        // is a single (kh,kw) and kw is wave-uniform, so the 8-pixel
        // window [kw, kw+8) can be cut out of 12 ALIGNED elements with
        // v_alignbyte instead of 8 scalar ds_read_u16 gathers.
        const bf16_t* arow0 = arow - kw;
        for (int w = 0; w < W; ++w) {
          const int pix0 = w * 32 + q * 8;
          bf16x8 afrag = (bf16x8)0;
          if (valid) {
            if (CI >= 16) {
              // aligned: row base is 16B-aligned (INPP*2 % 16 == 0) and
              // pix0*2 % 16 == 0
              const uint4 lo =
                  *reinterpret_cast<const uint4*>(arow0 + pix0);
              const uint2 hi =
                  *reinterpret_cast<const uint2*>(arow0 + pix0 + 8);
              // constant indices only (a runtime-indexed local array
              // would spill to scratch — guide rule 20); kw is a
              // wave-uniform scalar so the switch is one cheap branch
              const unsigned d0 = lo.x, d1 = lo.y, d2 = lo.z, d3 = lo.w;
              const unsigned d4 = hi.x, d5 = hi.y;
              unsigned o0, o1, o2, o3;
              switch (kw) {
                case 0: o0 = d0; o1 = d1; o2 = d2; o3 = d3; break;
                case 1:
                  o0 = __builtin_amdgcn_alignbyte(d1, d0, 2);
                  o1 = __builtin_amdgcn_alignbyte(d2, d1, 2);
                  o2 = __builtin_amdgcn_alignbyte(d3, d2, 2);
                  o3 = __builtin_amdgcn_alignbyte(d4, d3, 2);
                  break;
                case 2: o0 = d1; o1 = d2; o2 = d3; o3 = d4; break;
                case 3:
                  o0 = __builtin_amdgcn_alignbyte(d2, d1, 2);
                  o1 = __builtin_amdgcn_alignbyte(d3, d2, 2);
                  o2 = __builtin_amdgcn_alignbyte(d4, d3, 2);
                  o3 = __builtin_amdgcn_alignbyte(d5, d4, 2);
                  break;
                default: o0 = d2; o1 = d3; o2 = d4; o3 = d5; break;
              }
              union { uint4 u; bf16x8 h; } cv;
              cv.u = make_uint4(o0, o1, o2, o3);
              afrag = cv.h;
            } else {
              bf16_t av[8];
#pragma unroll
              for (int jj = 0; jj < 8; ++jj) av[jj] = arow[pix0 + jj];
              afrag = *reinterpret_cast<bf16x8*>(av);
            }
          }
          // B: lane l -> col o = l&15, pixels pix0..pix0+7
          const bf16x8 bfrag =
              *reinterpret_cast<const bf16x8*>(brow + pix0);
          acc[j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag,
                                                           acc[j], 0, 0, 0);
        }
      }
    }
    __syncthreads();  // before the next block restages lds_in/lds_gout
  }

  // ---- flush: part[wg][tap][o] (unique per wg -> plain stores)
  float* base = part + (long long)wg * T16 * CO;
#pragma unroll
  for (int j = 0; j < MAXP; ++j) {
    const int pair = wid + 4 * j;
    if (pair >= NPAIR) continue;
    const int tt = pair / COT;
    const int ot = pair - tt * COT;
    const int o = ot * 16 + (lane & 15);
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int tap = tt * 16 + q * 4 + i;
      base[(long long)tap * CO + o] = acc[j][i];
    }
  }
}

extern "C" {

// returns number of partial slabs written (grid size), or -1
int geops_conv5_wrw_nhwc(const bf16_t* in, const bf16_t* gout, float* part,
                         int Nn, int Hi, int Wi, int Ho, int Wo, int CI,
                         int CO, int n_wg, hipStream_t s) {
  const int W = (Wo + 31) >> 5;
  const int need = (W * 32 + 4) > Wi ? (W * 32 + 4) : Wi;
#define WLAUNCH(CI_, COT_, PIX_)                                         \
  if (need <= PIX_) {                                                    \
    hipLaunchKernelGGL((k_conv5_wrw_nhwc<CI_, COT_, PIX_>), dim3(n_wg),  \
                       dim3(WRW_THREADS), 0, s, in, gout, part, Nn, Hi,  \
                       Wi, Ho, Wo);                                      \
    return n_wg;                                                         \
  }
  if (CI == 4 && CO == 16) { WLAUNCH(4, 1, 136) WLAUNCH(4, 1, 232) }
  if (CI == 16 && CO == 32) { WLAUNCH(16, 2, 136) }
  if (CI == 16 && CO == 16) { WLAUNCH(16, 1, 136) WLAUNCH(16, 1, 232) }
#undef WLAUNCH
  return -1;
}

}  // extern "C"
