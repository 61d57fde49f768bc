// Direct 5x5 stride-1 convolution on MFMA for gfx950 — the flagship
// CNN's conv shapes (3->16 and 16->32 on 224/110 px, plus the padded
// transpose form for the data gradient), where MIOpen's igemm kernels
// run far from the memory floor.
//
// Formulation: implicit-GEMM. M = N*Ho*Wo output pixels, N = CO,
// K = 5 rows x Sp, Sp = round_up(5*CI, 8) elements per tap row
// ("row span"; NHWC makes (kw, ci) contiguous in memory, so every
// 8-element K-run of an MFMA A-fragment is ONE 16-byte load).
//
//   mfma_f32_16x16x32_bf16 per tile: 16 pixels x 16 outputs, K = 32.
//   A fragment: lane l = (q<<4)|p holds in[pixel p][k = km*32+q*8 .. +8]
//   B fragment: weights, PRE-PACKED on the host in exact per-lane
//     fragment order (w_frags[co_t][km][lane][8 bf16]) and loaded into
//     VGPRs once per wave for the whole grid-stride loop.
//   C/D: lane l holds out[pixel (l>>4)*4+i][o = l&15], i = 0..3.
//
// Border tiles (PAD > 0, used by the data-gradient pass which is a
// full correlation with spatially-flipped, CI<->CO-swapped weights)
// take a masked scalar-gather path; interior tiles (the vast majority)
// take the vector path.
//
// Requirements: CI % 4 == 0 (conv1's 3 channels are zero-padded to 4
// by the Python wrapper), CO % 16 == 0, kernel 5x5, stride 1.

#include <hip/hip_runtime.h>

#include <cstdint>

typedef unsigned short bf16_t;
typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define CONV_THREADS 256  // 4 waves

__device__ __forceinline__ float cbf2f(bf16_t h) {
  union { unsigned int u; float f; } cv;
  cv.u = ((unsigned int)h) << 16;
  return cv.f;
}

__device__ __forceinline__ bf16_t cf2bf(float f) {
  union { float f; unsigned int u; } cv;
  cv.f = f;
  // round-to-nearest-even
  unsigned int lsb = (cv.u >> 16) & 1;
  return (bf16_t)((cv.u + 0x7FFFu + lsb) >> 16);
}

// CI: input channels (mult of 4); COT: number of 16-wide output tiles;
// PAD: spatial zero padding (0 for fwd, 4 for the data-grad pass).
template <int CI, int COT, int PAD>
__global__ __launch_bounds__(CONV_THREADS) void k_conv5_nhwc(
    const bf16_t* __restrict__ in,       // [N][Hi][Wi][CI]
    const bf16_t* __restrict__ w_frags,  // [COT][nK][64][8]
    const float* __restrict__ bias,      // [CO] or nullptr
    bf16_t* __restrict__ out,            // [N][Ho][Wo][CO]
    int Nn, int Hi, int Wi, int Ho, int Wo) {
  constexpr int S = 5 * CI;                   // real row span
  constexpr int Sp = (S + 7) & ~7;            // padded row span
  constexpr int K = 5 * Sp;
  constexpr int nK = (K + 31) / 32;
  constexpr int CO = COT * 16;

  const int lane = threadIdx.x & 63;
  const int p = lane & 15;        // pixel slot within the tile
  const int q = lane >> 4;        // k-run selector

  // load B fragments once (they live in VGPRs across the whole loop)
  bf16x8 bfrag[COT][nK];
#pragma unroll
  for (int ct = 0; ct < COT; ++ct)
#pragma unroll
    for (int km = 0; km < nK; ++km)
      bfrag[ct][km] = *reinterpret_cast<const bf16x8*>(
          w_frags + (((long long)ct * nK + km) * 64 + lane) * 8);

  const int tiles_w = (Wo + 15) >> 4;
  const long long n_tiles = (long long)Nn * Ho * tiles_w;
  const long long wave_id =
      ((long long)blockIdx.x * blockDim.x + threadIdx.x) >> 6;
  const long long n_waves = ((long long)gridDim.x * blockDim.x) >> 6;

  for (long long t = wave_id; t < n_tiles; t += n_waves) {
    const int tw = (int)(t % tiles_w);
    long long rest = t / tiles_w;
    const int ho = (int)(rest % Ho);
    const long long n = rest / Ho;
    const int wo0 = tw << 4;
    const int wo_raw = wo0 + p;
    // clamp out-of-tile lanes to a valid pixel: their A rows feed only
    // D rows that the store masks out, so any in-bounds value is fine
    const int wo = wo_raw < Wo ? wo_raw : (Wo - 1);
    const int woM = (wo0 + 15) < Wo ? (wo0 + 15) : (Wo - 1);

    f32x4 acc[COT];
#pragma unroll
    for (int ct = 0; ct < COT; ++ct) acc[ct] = (f32x4)0.0f;

    // interior test (wave-uniform): every lane's loads in-bounds
    const bool interior = (ho >= PAD) && (ho - PAD + 4 < Hi) &&
                          (wo0 >= PAD) && (woM - PAD + 5 <= Wi);

    const long long in_n = (long long)n * Hi * Wi * CI;

#pragma unroll
    for (int km = 0; km < nK; ++km) {
      const int k0 = km * 32 + q * 8;
      const int kh = k0 / Sp;
      const int j0 = k0 % Sp;
      bf16x8 afrag = (bf16x8)0;
      const int hi = ho - PAD + kh;
      const bool k_valid = (k0 < K) && (j0 < S) && (kh < 5);
      if (k_valid && hi >= 0 && hi < Hi) {
        const int wbase = wo - PAD;  // elem column start for kw=0
        if (interior) {
          // vector path: one 16B (or 8B+zero) load
          const long long e = in_n + ((long long)hi * Wi + wbase) * CI + j0;
          if (S - j0 >= 8) {
            afrag = *reinterpret_cast<const bf16x8*>(in + e);
          } else {  // 4 real elements + 4 zero-pad (conv1's Sp > S)
            const uint2 v = *reinterpret_cast<const uint2*>(in + e);
            union { uint4 u; bf16x8 h; } cv;
            cv.u = make_uint4(v.x, v.y, 0u, 0u);
            afrag = cv.h;
          }
        } else {
          // masked scalar gather (border tiles of the padded pass)
          bf16_t tmp[8];
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            const int jj = j0 + j;
            const int kw = jj / CI;
            const int ci = jj - kw * CI;
            const int wi = wbase + kw;
            tmp[j] = (jj < S && wi >= 0 && wi < Wi)
                         ? in[in_n + ((long long)hi * Wi + wi) * CI + ci]
                         : (bf16_t)0;
          }
          afrag = *reinterpret_cast<bf16x8*>(tmp);
        }
      }
#pragma unroll
      for (int ct = 0; ct < COT; ++ct)
        acc[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag[ct][km], acc[ct], 0, 0, 0);
    }

    // store: lane l covers pixels (l>>4)*4+i at output channel l&15
    const int o = lane & 15;
#pragma unroll
    for (int ct = 0; ct < COT; ++ct) {
      const int oc = ct * 16 + o;
      const float bv = bias ? bias[oc] : 0.0f;
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        const int prow = q * 4 + i;
        const int w_out = wo0 + prow;
        if (w_out < Wo) {
          out[(((long long)n * Ho + ho) * Wo + w_out) * CO + oc] =
              cf2bf(acc[ct][i] + bv);
        }
      }
    }
  }
}

static inline int conv_blocks(long long tiles) {
  long long waves_needed = tiles;
  long long blocks = (waves_needed + 3) / 4;  // 4 waves per block
  if (blocks < 1) blocks = 1;
  if (blocks > 2048) blocks = 2048;
  return (int)blocks;
}

extern "C" {

// returns 0 on success, -1 for unsupported geometry
int geops_conv5_nhwc(const bf16_t* in, const bf16_t* w_frags,
                     const float* bias, bf16_t* out, int Nn, int Hi, int Wi,
                     int Ho, int Wo, int CI, int CO, int pad,
                     hipStream_t s) {
  const long long tiles = (long long)Nn * Ho * ((Wo + 15) >> 4);
  const dim3 grid(conv_blocks(tiles)), block(CONV_THREADS);
#define LAUNCH(CI_, COT_, PAD_)                                         \
  hipLaunchKernelGGL((k_conv5_nhwc<CI_, COT_, PAD_>), grid, block, 0, s, \
                     in, w_frags, bias, out, Nn, Hi, Wi, Ho, Wo);        \
  return 0;
  if (pad == 0) {
    if (CI == 4 && CO == 16) { LAUNCH(4, 1, 0) }
    if (CI == 16 && CO == 32) { LAUNCH(16, 2, 0) }
    if (CI == 16 && CO == 16) { LAUNCH(16, 1, 0) }
    if (CI == 32 && CO == 32) { LAUNCH(32, 2, 0) }
  } else if (pad == 4) {
    if (CI == 32 && CO == 16) { LAUNCH(32, 1, 4) }
    if (CI == 16 && CO == 16) { LAUNCH(16, 1, 4) }
    if (CI == 16 && CO == 4) { /* unsupported CO<16 */ }
  }
#undef LAUNCH
  return -1;
}

}  // extern "C"
