// MFMA weight-gradient (wrw) kernel v2 for gfx950 — tr_b16 staging.
//
// dW[k, (r,s,c)] = sum_{pix} gy[pix, k] * x[src(pix,r,s), c]: a GEMM
// whose reduction axis (pixels) is the OUTER dimension of both NHWC
// operands.  v1 transposed while staging with 16 scalar bf16 LDS
// writes per thread per pixel-pair (measured 0.25-0.9x MIOpen).  v2
// keeps the NATURAL [pix][ch] orientation in LDS — one bf16x8 global
// load + one ds_write_b128 per (pixel, 8-ch chunk) — and does the
// transpose at CONSUME time with gfx950's hardware transpose read
// (ds_read_b64_tr_b16): each MFMA fragment is two tr reads from a
// [ks][pix][16] subtile image (the guide's conflict-free
// lds[(l&15)+j*16+(l>>4)*64] layout, subtiles padded 32 B so the
// b128 write pass covers all 32 banks).
//
// Pixel range split over grid.y; fp32 atomic accumulation into dW
// (output is K x RSC — tiny next to the reduction reads).

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.hip.h"

static inline hipStream_t deepof_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

namespace {

using bf16 = __bf16;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short short8v;
typedef __attribute__((address_space(3))) bf16x4 lds_bf16x4;

// ks-subtile stride in elements: [BKP][16] image + 16-elem pad so the
// 8-lane ds_write_b128 groups of the staging pass land on 32 distinct
// banks (ks advances banks by 8, the k0&15 in-subtile offset by 4)
template <int BKP>
constexpr int SS = BKP * 16 + 16;

// BKP = pixels per stage (the MFMA reduction tile)
template <int BKP>
__global__ __launch_bounds__(256)
void conv_wrw2_kernel(const bf16* __restrict__ gy,  // [B,OH,OW,K]
                      const bf16* __restrict__ x,   // [B,IH,IW,C]
                      float* __restrict__ dw,       // [K, RSC] fp32
                      int B, int IH, int IW, int C, int K,
                      int R, int S, int OH, int OW,
                      int stride, int pad,
                      int n_tiles_n, int pix_per_slice) {
  // one __shared__ object: [2 dbuf][2 op][4 ks][SS]
  __shared__ bf16 lds_all[2 * 2 * 4 * SS<BKP>];
#define W2_LDS(buf, op) (lds_all + ((buf) * 2 + (op)) * 4 * SS<BKP>)

  const int tile_k = (blockIdx.x / n_tiles_n) * 64;   // K offset
  const int tile_n = (blockIdx.x % n_tiles_n) * 64;   // RSC offset
  const int M = B * OH * OW;
  const int pix0 = blockIdx.y * pix_per_slice;
  const int pix_end = min(pix0 + pix_per_slice, M);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;            // 4 waves, 2x2
  const int wm = (wid >> 1) * 32;      // wave K offset within tile
  const int wn = (wid & 1) * 32;       // wave RSC offset

  // the 64-wide rsc tile stays inside one (r,s) tap (host: C % 64 == 0)
  const int rs = tile_n / C;
  const int rr = rs / S, ss = rs % S;
  const int c0 = tile_n % C;

  // ---- staging geometry: thread = (pixel p, 8-ch chunk k0) ----
  const int st_p = tid >> 3;                 // 0..31 (2 passes for BKP=64)
  const int st_k0 = (tid & 7) * 8;           // channel chunk base
  const int st_ks = st_k0 >> 4;              // ks subtile
  const int st_ko = st_k0 & 15;              // offset inside subtile row
  constexpr int P_STEP = 32;                 // pixels per pass

  auto stage = [&](int pix_base, int buf) {
#pragma unroll
    for (int pass = 0; pass < BKP / P_STEP; ++pass) {
      const int pl = st_p + pass * P_STEP;   // pixel within stage
      const int p = pix_base + pl;
      // gy tile: rows = k
      bf16x8 a = {};
      if (p < pix_end && tile_k + st_k0 < K)
        a = *reinterpret_cast<const bf16x8*>(
            gy + (long)p * K + tile_k + st_k0);
      // x tile: rows = c (within the tap)
      bf16x8 b = {};
      if (p < pix_end) {
        const int ox = p % OW;
        const int oy = (p / OW) % OH;
        const int bb = p / (OW * OH);
        const int iy = oy * stride + rr - pad;
        const int ix = ox * stride + ss - pad;
        if (iy >= 0 && iy < IH && ix >= 0 && ix < IW)
          b = *reinterpret_cast<const bf16x8*>(
              x + (((long)bb * IH + iy) * IW + ix) * C + c0 + st_k0);
      }
      const int dst = st_ks * SS<BKP> + pl * 16 + st_ko;
      *reinterpret_cast<bf16x8*>(&W2_LDS(buf, 0)[dst]) = a;
      *reinterpret_cast<bf16x8*>(&W2_LDS(buf, 1)[dst]) = b;
    }
  };

  // tr-read one 8-pixel fragment: rows ch0..ch0+15 (one ks subtile),
  // reduction pixels p0..p0+7.  Measured ds_read_b64_tr_b16 semantics
  // (tools/tr_probe.hip): each lane supplies an 8-B-aligned address of
  // a 4-element bf16 run; the result is the cross-lane transpose
  //   out[l][j] = lds[addr_of_lane(16*(l>>4) + 4*j + ((l&15)>>2))
  //                   + (l&3)].
  // With per-lane address (elements)
  //   ks*SS + (p0 + 8*(l>>4) + ((l>>2)&3))*16 + 4*(l&3)
  // into the [pix][16ch] image, lane l elem j delivers
  // A[ch0 + (l&15)][p0 + 8*(l>>4) + j] — exactly the 16x16x32 MFMA
  // operand layout; the +64-element read supplies j = 4..7.
  auto tr_frag = [&](const bf16* img, int ch0, int p0) -> bf16x8 {
    const int ks = ch0 >> 4;
    const bf16* base = img + ks * SS<BKP> +
                       (p0 + 8 * (lane >> 4) + ((lane >> 2) & 3)) * 16 +
                       4 * (lane & 3);
    bf16x4 lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (lds_bf16x4*)base);
    bf16x4 hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
        (lds_bf16x4*)(base + 64));
    bf16x8 out;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      out[j] = lo[j];
      out[j + 4] = hi[j];
    }
    return out;
  };

  f32x4 acc[2][2] = {};
  const int n_stages = (pix_end - pix0 + BKP - 1) / BKP;
  if (n_stages <= 0) return;

  stage(pix0, 0);
  __syncthreads();
  for (int st = 0; st < n_stages; ++st) {
    const int buf = st & 1;
    if (st + 1 < n_stages) stage(pix0 + (st + 1) * BKP, buf ^ 1);
#pragma unroll
    for (int kk = 0; kk < BKP; kk += 32) {
      bf16x8 afrag[2], bfrag[2];
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
        afrag[mi] = tr_frag(W2_LDS(buf, 0), wm + mi * 16, kk);
#pragma unroll
      for (int ni = 0; ni < 2; ++ni)
        bfrag[ni] = tr_frag(W2_LDS(buf, 1), wn + ni * 16, kk);
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
#pragma unroll
        for (int ni = 0; ni < 2; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              __builtin_bit_cast(short8v, afrag[mi]),
              __builtin_bit_cast(short8v, bfrag[ni]), acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: atomic fp32 accumulation into dW
  const int RSC = R * S * C;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 2; ++ni) {
      const int n = tile_n + wn + ni * 16 + (lane & 15);
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int k = tile_k + wm + mi * 16 + (lane >> 4) * 4 + reg;
        if (k < K)
          atomicAdd(dw + (long)k * RSC + n, acc[mi][ni][reg]);
      }
    }
  }
#undef W2_LDS
}

}  // namespace

// gy [B,K,OH,OW] channels_last, x [B,C,IH,IW] channels_last ->
// dW [K,C,R,S] channels_last bf16
at::Tensor conv2d_wrw2(at::Tensor gy, at::Tensor x, long R_, long S_,
                       long stride, long pad) {
  TORCH_CHECK(gy.is_cuda() && x.is_cuda());
  TORCH_CHECK(gy.scalar_type() == at::kBFloat16 &&
              x.scalar_type() == at::kBFloat16);
  TORCH_CHECK(gy.is_contiguous(at::MemoryFormat::ChannelsLast) &&
              x.is_contiguous(at::MemoryFormat::ChannelsLast));
  const int B = x.size(0), C = x.size(1), IH = x.size(2), IW = x.size(3);
  const int K = gy.size(1), OH = gy.size(2), OW = gy.size(3);
  const int R = (int)R_, S = (int)S_;
  TORCH_CHECK(C % 64 == 0, "conv2d_wrw2 needs C % 64 == 0");
  TORCH_CHECK(K % 8 == 0);
  const int RSC = R * S * C;
  const long M = (long)B * OH * OW;

  auto dw_f32 = at::zeros({K, RSC}, x.options().dtype(at::kFloat));
  const int ktiles = (K + 63) / 64;
  const int ntiles = RSC / 64;
  static const int BKP = [] {
    const char* e = getenv("DEEPOF_WRW2_BKP");
    return e ? atoi(e) : 128;
  }();
  // pixel split so the grid fills the chip (~2 blocks/CU)
  int split = (int)std::max(1L, 512L / ((long)ktiles * ntiles));
  const int pix_per_slice =
      (int)(((M + split - 1) / split + BKP - 1) / BKP) * BKP;
  split = (int)((M + pix_per_slice - 1) / pix_per_slice);

  const dim3 grid(ktiles * ntiles, split), block(256);
#define WRW2_LAUNCH(BKP_)                                                  \
  hipLaunchKernelGGL((conv_wrw2_kernel<BKP_>), grid, block, 0,             \
                     deepof_stream(),                                      \
                     reinterpret_cast<const bf16*>(gy.data_ptr()),         \
                     reinterpret_cast<const bf16*>(x.data_ptr()),          \
                     dw_f32.data_ptr<float>(), B, IH, IW, C, K, R, S,      \
                     OH, OW, (int)stride, (int)pad, ntiles,                \
                     pix_per_slice)
  if (BKP == 128) WRW2_LAUNCH(128); else WRW2_LAUNCH(64);
#undef WRW2_LAUNCH
  auto dw = dw_f32.view({K, R, S, C}).permute({0, 3, 1, 2}).to(at::kBFloat16);
  return dw.contiguous(at::MemoryFormat::ChannelsLast);
}
