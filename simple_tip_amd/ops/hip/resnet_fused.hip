// Fused ResNet-20 inference kernels for gfx950 (bf16 NHWC, CDNA4 MFMA).
//
// Motivation (profiles/r01_bench_kernels.md): MIOpen runs the bench's
// AT-extraction forward at ~26 TF effective — per-layer launches, im2col
// staging and BatchNorm inference dominate. These kernels compute one
// RESIDUAL BLOCK per launch: the input plane (with 1-pixel halo) is staged
// into LDS once, conv1 -> relu writes the intermediate plane to LDS only,
// conv2 + bias + residual-add + relu writes the block output to HBM. HBM
// traffic per block = read input + write output; the intermediate never
// leaves the CU.
//
// Layout/geometry:
// - activations: NHWC bf16; LDS image [(H+2) x (W+2) x C] with zeroed halo,
//   16-byte units XOR-swizzled (u ^= (u>>4)&7) so the per-tap
//   ds_read_b128 gathers are bank-conflict-free for C in {16, 32, 64}.
// - MFMA: v_mfma_f32_16x16x32_bf16. M = 16 consecutive output pixels
//   (row-major in the plane), N = 16 output channels, K = 32 consecutive
//   (tap, channel) pairs of the 3x3xC patch (zero-padded to a multiple
//   of 32). A-fragment: lane l holds A[i = l&15][k = (l>>4)*8 + e]
//   (one 16-B LDS read per lane); B-fragment comes pre-packed on the host
//   into [ksteps][64][8] bf16 so each lane reads its 16 B straight from
//   global (L2-resident; weights are <= 73 KB per layer).
//   C/D: col = lane&15, row = (lane>>4)*4 + reg.
//   (The layout is verified at runtime by the mfma_probe test kernel.)
// - one workgroup = one image; 4 waves split the plane's 16-pixel tiles.

#include "tip_common.h"

#include <hip/hip_bf16.h>

using bf16 = __hip_bfloat16;
using short8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

// 16-B-unit XOR swizzle family: u ^= ((u>>4) * M) & 15 (injective: only
// bits 0-3 change, driven by bits >= 4). M = 0 (identity, default) won a
// 9-variant hardware sweep by +22% (364 vs ~300 TF on resblock<32,32,16>):
// the ds_read_b128 lane-service groups are non-contiguous, and for this
// access pattern the linear image's residual conflicts cost less than the
// per-access XOR address math (profiles/r01_optimization_ladder.md).
#ifndef TIP_SWZ_M
#define TIP_SWZ_M 0
#endif
TIP_DEV int swz(int u) {
  if (TIP_SWZ_M == 0) return u;
  return u ^ (((u >> 4) * TIP_SWZ_M) & 15);
}

// LDS image layout. Round-1 PMC runs measured residual bank conflicts on
// the per-tap A-fragment gathers (profiles/r01_optimization_ladder.md) and
// sketched a "half-channel plane" layout (C/8 planes of one 16-B unit per
// pixel, making same-row gathers stride-1) as the fix. MEASURED IN ROUND 2
// AND REJECTED: on hardware the half-plane layout RAISES
// SQ_LDS_BANK_CONFLICT 3.2x (1.68e7 -> 5.45e7 per 3-launch probe) and
// costs 8% (382 -> 353 TF on resblock<32,32,16>) — the b128 service
// grouping evidently pairs lanes across k-groups, where the packed
// layout's (tap, half-channel) offsets already land in distinct bank
// groups and the stride-C/8 pixel walk does not collide the way a
// same-row-lanes model predicts. The packed unit-row-major image
// (TIP_HALFPLANE=0) stays the default; the half-plane variant is kept
// compilable for future counter runs.
#ifndef TIP_HALFPLANE
#define TIP_HALFPLANE 0
#endif

template <int H, int W, int C>
struct Img {
  static constexpr int UPP = C / 8;  // 16-B units per pixel
#if TIP_HALFPLANE
  static constexpr int RS1 = W + 2;          // units per halo row per plane
  static constexpr int PLANE = (H + 2) * RS1;
  static constexpr int UNITS = UPP * PLANE;
  static constexpr int XSTEP = 1;            // unit step per +1 output x
  static TIP_DEV int unit(int y, int x, int h) {
    return h * PLANE + y * RS1 + x;
  }
#else
  static constexpr int UNITS = (H + 2) * (W + 2) * UPP;
  static constexpr int XSTEP = UPP;
  static TIP_DEV int unit(int y, int x, int h) {
    return (y * (W + 2) + x) * UPP + h;
  }
#endif
};

// Load a 16-B unit (8 bf16) from the swizzled LDS image.
TIP_DEV short8 lds_read_unit(const short* lds, int u) {
  return *reinterpret_cast<const short8*>(lds + swz(u) * 8);
}

TIP_DEV void lds_write_unit(short* lds, int u, short8 v) {
  *reinterpret_cast<short8*>(lds + swz(u) * 8) = v;
}

// Stage one NHWC plane [H x W x C] from global into the LDS image
// [(H+2) x (W+2) x C] interior; halo is zeroed first. 256 threads.
template <int H, int W, int C>
TIP_DEV void stage_plane(short* lds, const short* __restrict__ gsrc) {
  using L = Img<H, W, C>;
  const short8 zero = {0, 0, 0, 0, 0, 0, 0, 0};
  for (int u = threadIdx.x; u < L::UNITS; u += blockDim.x)
    lds_write_unit(lds, u, zero);
  __syncthreads();
  constexpr int UNITS = H * W * C / 8;
  constexpr int UPP = C / 8;  // units per pixel
  for (int u = threadIdx.x; u < UNITS; u += blockDim.x) {
    const int p = u / UPP;
    const int h = u - p * UPP;
    const int y = p / W, x = p - y * W;
    const int lu = L::unit(y + 1, x + 1, h);
    short8 v = *reinterpret_cast<const short8*>(gsrc + (int64_t)u * 8);
    lds_write_unit(lds, lu, v);
  }
}

// 3x3 conv over the LDS image: computes the [16-pixel x 16-cout] tiles
// this wave owns and applies bias (+ optional residual from rlds) + relu,
// then writes either to the out LDS image interior or to global NHWC.
// K layout: k = tap*C + ci, taps row-major dy,dx in [0,3)x[0,3).
template <int H, int W, int C, int COUT, bool TO_LDS, bool RESID, int STRIDE>
TIP_DEV void conv3x3(
    const short* __restrict__ in_lds,   // [(H+2)(W+2)C] swizzled
    short* __restrict__ out_lds,        // TO_LDS: [(OH+2)(OW+2)COUT]
    short* __restrict__ gout,           // !TO_LDS: global NHWC [OH*OW*COUT]
    const short* __restrict__ rlds,     // RESID: residual LDS image (COUT ch)
    const short* __restrict__ wpack,    // [ksteps][64][8] per cout-tile:
                                        // [cout_tiles][ksteps][64][8]
    const float* __restrict__ bias) {   // [COUT]
  constexpr int OH = H / STRIDE, OW = W / STRIDE;
  constexpr int K = 9 * C;
  constexpr int KSTEPS = (K + 31) / 32;
  constexpr int NPIX = OH * OW;
  constexpr int PIX_TILES = NPIX / 16;
  constexpr int COUT_TILES = COUT / 16;
  using In = Img<H, W, C>;
  using Out = Img<OH, OW, COUT>;

  const int lane = lane_id();
  const int wid = wave_id();
  const int j = lane & 15;        // output channel within tile
  const int g = lane >> 4;        // k-group (8 consecutive k)

  // Small-plane specialization (PIX_TILES <= 4, i.e. the 8x8 stage): each
  // wave owns at most ONE pixel tile, so the generic path's dual-PIXEL
  // MFMA chains collapse to a single dependent chain (measured 24% slower
  // per FLOP than the 32x32 stage). Interleave TWO COUT tiles instead:
  // the A fragment is shared, the B fragments load from L2 per k-step
  // (high occupancy here — 6 blocks/CU — hides the load latency the
  // preload design was protecting the low-occupancy stages from). The
  // per-(ct, ks) accumulation order is unchanged, so results are bitwise
  // identical to the generic path.
  if constexpr (PIX_TILES <= 4 && (COUT_TILES % 2) == 0) {
    const int pt = wid;
    if (pt >= PIX_TILES) return;
    const int apix = pt * 16 + j;
    const int aoy = apix / OW, aox = apix - aoy * OW;
    const int abase = In::unit(aoy * STRIDE, aox * STRIDE, 0);
    for (int ct = 0; ct < COUT_TILES; ct += 2) {
      const short* wp0 = wpack + ((int64_t)ct * KSTEPS) * 64 * 8 + lane * 8;
      const short* wp1 = wp0 + (int64_t)KSTEPS * 64 * 8;
      f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
      f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) {
        const int k0 = ks * 32 + g * 8;
        short8 a;
        if (k0 < K) {
          const int tap = k0 / C;
          const int ci = k0 & (C - 1);
          const int dy = (tap * 11) >> 5;
          const int dx = tap - dy * 3;
          a = lds_read_unit(in_lds, abase + In::unit(dy, dx, ci >> 3));
        } else {
          a = short8{0, 0, 0, 0, 0, 0, 0, 0};
        }
        const short8 b0 =
            *reinterpret_cast<const short8*>(wp0 + (int64_t)ks * 64 * 8);
        const short8 b1 =
            *reinterpret_cast<const short8*>(wp1 + (int64_t)ks * 64 * 8);
        acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b0, acc0, 0, 0, 0);
        acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b1, acc1, 0, 0, 0);
      }
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        const int cout = (ct + half) * 16 + j;
        const float bs = bias[cout];
        const f32x4& acc = half ? acc1 : acc0;
        const int pix0 = pt * 16 + g * 4;
        const int oy = pix0 / OW, ox = pix0 - oy * OW;
        const int u0 = Out::unit(oy + 1, ox + 1, cout >> 3);
        const int e = cout & 7;
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          float v = acc[reg] + bs;
          if (RESID) {
            const short* runit = rlds + swz(u0 + reg * Out::XSTEP) * 8;
            v += __bfloat162float(reinterpret_cast<const bf16*>(runit)[e]);
          }
          v = fmaxf(v, 0.f);
          const bf16 ov = __float2bfloat16(v);
          if (TO_LDS) {
            short* unit = out_lds + swz(u0 + reg * Out::XSTEP) * 8;
            reinterpret_cast<bf16*>(unit)[e] = ov;
          } else {
            reinterpret_cast<bf16*>(gout)[(int64_t)(pix0 + reg) * COUT + cout] =
                ov;
          }
        }
      }
    }
    return;
  }

  // cout-tile outer loop: the tile's weight fragments are preloaded into
  // registers ONCE (KSTEPS x 16 B per lane) so the MFMA loop is pure
  // ds_read + mfma — no global loads on the critical path (the per-mfma
  // L2 weight load was the previous bound at ~2 waves/SIMD occupancy).
  for (int ct = 0; ct < COUT_TILES; ++ct) {
    short8 bfrag[KSTEPS];
    {
      const short* wp = wpack + ((int64_t)ct * KSTEPS) * 64 * 8 + lane * 8;
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks)
        bfrag[ks] = *reinterpret_cast<const short8*>(wp + (int64_t)ks * 64 * 8);
    }
    const int cout = ct * 16 + j;
    const float bs = bias[cout];

    // LDS unit offset of k-step ks relative to a pixel's top-left tap
    // (constexpr after full unroll): k decomposes into (tap, channel) with
    // power-of-two C, dy = tap/3 via mul-shift (tap < 10).
    auto a_off = [&](int ks) {
      const int k0 = ks * 32 + g * 8;
      const int tap = k0 / C;
      const int ci = k0 & (C - 1);
      const int dy = (tap * 11) >> 5;
      const int dx = tap - dy * 3;
      return In::unit(dy, dx, ci >> 3);
    };
    auto a_base = [&](int pt) {
      const int apix = pt * 16 + j;
      const int aoy = apix / OW, aox = apix - aoy * OW;
      return In::unit(aoy * STRIDE, aox * STRIDE, 0);
    };
    auto epilogue = [&](int pt, const f32x4& acc) {
      // the 4 regs of a C/D fragment are 4 consecutive pixels in ONE output
      // row (4 <= OW always), so the halo-image unit address is linear in
      // reg: one div/mod per fragment, +XSTEP per register
      const int pix0 = pt * 16 + g * 4;
      const int oy = pix0 / OW, ox = pix0 - oy * OW;
      const int u0 = Out::unit(oy + 1, ox + 1, cout >> 3);
      const int e = cout & 7;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float v = acc[reg] + bs;
        if (RESID) {
          const short* runit = rlds + swz(u0 + reg * Out::XSTEP) * 8;
          v += __bfloat162float(reinterpret_cast<const bf16*>(runit)[e]);
        }
        v = fmaxf(v, 0.f);
        const bf16 ov = __float2bfloat16(v);
        if (TO_LDS) {
          short* unit = out_lds + swz(u0 + reg * Out::XSTEP) * 8;
          reinterpret_cast<bf16*>(unit)[e] = ov;
        } else {
          reinterpret_cast<bf16*>(gout)[(int64_t)(pix0 + reg) * COUT + cout] = ov;
        }
      }
    };

    // Two independent pixel tiles per iteration: two MFMA accumulation
    // chains interleave, hiding the dependent-accumulator and LDS-read
    // latency that a single chain serialises on.
    int pt = wid;
    for (; pt + 4 < PIX_TILES; pt += 8) {
      const int b0 = a_base(pt), b1 = a_base(pt + 4);
      f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
      f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) {
        const int k0 = ks * 32 + g * 8;
        short8 a0, a1;
        if (k0 < K) {
          const int off = a_off(ks);
          a0 = lds_read_unit(in_lds, b0 + off);
          a1 = lds_read_unit(in_lds, b1 + off);
        } else {
          a0 = short8{0, 0, 0, 0, 0, 0, 0, 0};
          a1 = a0;
        }
        acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, bfrag[ks], acc0, 0, 0, 0);
        acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, bfrag[ks], acc1, 0, 0, 0);
      }
      epilogue(pt, acc0);
      epilogue(pt + 4, acc1);
    }
    for (; pt < PIX_TILES; pt += 4) {
      const int b0 = a_base(pt);
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < KSTEPS; ++ks) {
        const int k0 = ks * 32 + g * 8;
        short8 a;
        if (k0 < K) {
          a = lds_read_unit(in_lds, b0 + a_off(ks));
        } else {
          a = short8{0, 0, 0, 0, 0, 0, 0, 0};
        }
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bfrag[ks], acc, 0, 0, 0);
      }
      epilogue(pt, acc);
    }
  }
}

// 1x1 stride-2 shortcut projection, accumulated in the conv2 epilogue is
// complex; instead we compute the projected residual into rlds first.
// rlds: [(OH+2) x (OW+2) x COUT] swizzled image (interior only written;
// halo must be pre-zeroed). wpack1x1: K = C padded to 32.
template <int H, int W, int C, int COUT>
TIP_DEV void shortcut1x1_s2(
    const short* __restrict__ in_lds, short* __restrict__ rlds,
    const short* __restrict__ wpack, const float* __restrict__ bias) {
  constexpr int OH = H / 2, OW = W / 2;
  constexpr int KSTEPS = (C + 31) / 32;
  constexpr int NPIX = OH * OW;
  constexpr int PIX_TILES = NPIX / 16;
  constexpr int COUT_TILES = COUT / 16;
  using In = Img<H, W, C>;
  using Out = Img<OH, OW, COUT>;
  const int lane = lane_id();
  const int wid = wave_id();
  const int j = lane & 15;
  const int g = lane >> 4;
  for (int tile = wid; tile < PIX_TILES * COUT_TILES; tile += 4) {
    const int pt = tile % PIX_TILES;
    const int ct = tile / PIX_TILES;
    const int p0 = pt * 16;
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const short* wp = wpack + ((int64_t)ct * KSTEPS) * 64 * 8 + lane * 8;
    for (int ks = 0; ks < KSTEPS; ++ks) {
      const int k0 = ks * 32 + g * 8;
      short8 a = {0, 0, 0, 0, 0, 0, 0, 0};
      if (k0 < C) {
        const int pix = p0 + j;
        const int oy = pix / OW, ox = pix - oy * OW;
        const int iy = oy * 2 + 1, ix = ox * 2 + 1;  // center tap, halo coords
        a = lds_read_unit(in_lds, In::unit(iy, ix, k0 >> 3));
      }
      const short8 b = *reinterpret_cast<const short8*>(wp + (int64_t)ks * 64 * 8);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    const int cout = ct * 16 + j;
    const float bs = bias[cout];
    const int pix0 = p0 + g * 4;
    const int oy = pix0 / OW, ox = pix0 - oy * OW;
    const int u0 = Out::unit(oy + 1, ox + 1, cout >> 3);
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      short* unit = rlds + swz(u0 + reg * Out::XSTEP) * 8;
      reinterpret_cast<bf16*>(unit)[cout & 7] = __float2bfloat16(acc[reg] + bs);
    }
  }
}

// ---- kernels ----

// Equal-channel residual block: x -> conv1+relu (LDS) -> conv2+bias+x+relu
// -> global. One image per workgroup.
template <int H, int W, int C>
__launch_bounds__(256) __global__ void resblock_kernel(
    const short* __restrict__ gin,   // [B, H*W*C] NHWC bf16
    short* __restrict__ gout,        // [B, H*W*C]
    const short* __restrict__ w1, const float* __restrict__ b1,
    const short* __restrict__ w2, const float* __restrict__ b2) {
  extern __shared__ short lds[];
  short* bufX = lds;                                // image of Img<H,W,C>
  short* bufH = lds + Img<H, W, C>::UNITS * 8;      // second image
  const int64_t img_off = (int64_t)blockIdx.x * H * W * C;
  stage_plane<H, W, C>(bufX, gin + img_off);
  // zero bufH halo (stage_plane zeroes everything first; emulate)
  {
    const short8 zero = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int u = threadIdx.x; u < Img<H, W, C>::UNITS; u += blockDim.x)
      lds_write_unit(bufH, u, zero);
  }
  __syncthreads();
  conv3x3<H, W, C, C, true, false, 1>(bufX, bufH, nullptr, nullptr, w1, b1);
  __syncthreads();
  conv3x3<H, W, C, C, false, true, 1>(
      bufH, nullptr, gout + img_off, bufX, w2, b2);
}

// Downsample block: conv1 (stride 2, C -> 2C) + relu -> conv2 (2C) +
// shortcut(1x1 s2) + relu -> global.
template <int H, int W, int C>
__launch_bounds__(256) __global__ void downblock_kernel(
    const short* __restrict__ gin,   // [B, H*W*C]
    short* __restrict__ gout,        // [B, (H/2)*(W/2)*2C]
    const short* __restrict__ w1, const float* __restrict__ b1,
    const short* __restrict__ w2, const float* __restrict__ b2,
    const short* __restrict__ wsc, const float* __restrict__ bsc) {
  constexpr int OH = H / 2, OW = W / 2, C2 = 2 * C;
  extern __shared__ short lds[];
  short* bufX = lds;                                   // Img<H,W,C>
  short* bufH = bufX + Img<H, W, C>::UNITS * 8;        // Img<OH,OW,C2>
  short* bufR = bufH + Img<OH, OW, C2>::UNITS * 8;     // Img<OH,OW,C2>
  const int64_t in_off = (int64_t)blockIdx.x * H * W * C;
  const int64_t out_off = (int64_t)blockIdx.x * OH * OW * C2;
  stage_plane<H, W, C>(bufX, gin + in_off);
  {
    const short8 zero = {0, 0, 0, 0, 0, 0, 0, 0};
    for (int u = threadIdx.x; u < Img<OH, OW, C2>::UNITS; u += blockDim.x) {
      lds_write_unit(bufH, u, zero);
      lds_write_unit(bufR, u, zero);
    }
  }
  __syncthreads();
  conv3x3<H, W, C, C2, true, false, 2>(bufX, bufH, nullptr, nullptr, w1, b1);
  shortcut1x1_s2<H, W, C, C2>(bufX, bufR, wsc, bsc);
  __syncthreads();
  conv3x3<OH, OW, C2, C2, false, true, 1>(
      bufH, nullptr, gout + out_off, bufR, w2, b2);
}

// Stem: conv3x3 Cin=8 (4 real + 4 zero-pad channels; 8 keeps units whole)
// -> 16, relu, to global.
template <int H, int W, int CIN, int COUT>
__launch_bounds__(256) __global__ void stem_kernel(
    const short* __restrict__ gin,   // [B, H*W*CIN]
    short* __restrict__ gout,        // [B, H*W*COUT]
    const short* __restrict__ w, const float* __restrict__ b) {
  extern __shared__ short lds[];
  short* bufX = lds;
  const int64_t in_off = (int64_t)blockIdx.x * H * W * CIN;
  const int64_t out_off = (int64_t)blockIdx.x * H * W * COUT;
  stage_plane<H, W, CIN>(bufX, gin + in_off);
  __syncthreads();
  conv3x3<H, W, CIN, COUT, false, false, 1>(
      bufX, nullptr, gout + out_off, nullptr, w, b);
}

// MFMA layout probe: D = A @ B for A [16, 32], B [32, 16] bf16 (row-major)
// with the exact fragment code paths used above.
__global__ void mfma_probe_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    float* __restrict__ D) {
  const int lane = lane_id();
  const int j = lane & 15, g = lane >> 4;
  short8 a = *reinterpret_cast<const short8*>(A + (j * 32 + g * 8));
  short8 b;
  for (int e = 0; e < 8; ++e) b[e] = B[(g * 8 + e) * 16 + j];
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  for (int reg = 0; reg < 4; ++reg) D[(g * 4 + reg) * 16 + j] = acc[reg];
}

// ---- launchers ----

void launch_mfma_probe(const short* a, const short* b, float* d, hipStream_t s) {
  mfma_probe_kernel<<<1, 64, 0, s>>>(a, b, d);
}

template <int H, int W, int C>
static void resblock(int batch, const short* gin, short* gout, const short* w1,
                     const float* b1, const short* w2, const float* b2,
                     hipStream_t s) {
  const int lds_bytes = 2 * Img<H, W, C>::UNITS * 16;
  auto k = resblock_kernel<H, W, C>;
  hipFuncSetAttribute((const void*)k,
                      hipFuncAttributeMaxDynamicSharedMemorySize, lds_bytes);
  k<<<batch, 256, lds_bytes, s>>>(gin, gout, w1, b1, w2, b2);
}

void launch_resblock(int variant, int batch, const short* gin, short* gout,
                     const short* w1, const float* b1, const short* w2,
                     const float* b2, hipStream_t s) {
  if (variant == 0) resblock<32, 32, 16>(batch, gin, gout, w1, b1, w2, b2, s);
  else if (variant == 1) resblock<16, 16, 32>(batch, gin, gout, w1, b1, w2, b2, s);
  else resblock<8, 8, 64>(batch, gin, gout, w1, b1, w2, b2, s);
}

template <int H, int W, int C>
static void downblock(int batch, const short* gin, short* gout,
                      const short* w1, const float* b1, const short* w2,
                      const float* b2, const short* wsc, const float* bsc,
                      hipStream_t s) {
  constexpr int OH = H / 2, OW = W / 2, C2 = 2 * C;
  const int lds_bytes =
      (Img<H, W, C>::UNITS + 2 * Img<OH, OW, C2>::UNITS) * 16;
  auto k = downblock_kernel<H, W, C>;
  hipFuncSetAttribute((const void*)k,
                      hipFuncAttributeMaxDynamicSharedMemorySize, lds_bytes);
  k<<<batch, 256, lds_bytes, s>>>(gin, gout, w1, b1, w2, b2, wsc, bsc);
}

void launch_downblock(int variant, int batch, const short* gin, short* gout,
                      const short* w1, const float* b1, const short* w2,
                      const float* b2, const short* wsc, const float* bsc,
                      hipStream_t s) {
  if (variant == 0)
    downblock<32, 32, 16>(batch, gin, gout, w1, b1, w2, b2, wsc, bsc, s);
  else
    downblock<16, 16, 32>(batch, gin, gout, w1, b1, w2, b2, wsc, bsc, s);
}

void launch_stem(int batch, const short* gin, short* gout, const short* w,
                 const float* b, hipStream_t s) {
  constexpr int H = 32, W = 32, CIN = 8, COUT = 16;
  const int lds_bytes = Img<H, W, CIN>::UNITS * 16;
  auto k = stem_kernel<H, W, CIN, COUT>;
  hipFuncSetAttribute((const void*)k,
                      hipFuncAttributeMaxDynamicSharedMemorySize, lds_bytes);
  k<<<batch, 256, lds_bytes, s>>>(gin, gout, w, b);
}
