// MFMA weight-gradient (wrw) kernel for gfx950.
//
// dW[k, (r,s,c)] = sum_{pix in B*OH*OW} gy[pix, k] * x[src(pix,r,s), c]
// — a GEMM whose reduction axis (pixels) is the OUTER dimension of
// both NHWC operands, so both tiles are transposed while staging:
// bf16x8 coalesced loads (contiguous channels), pix-pair-packed
// ds_write_b32 into [k][pix] / [rsc][pix] LDS images, then the same
// swizzled ds_read_b128 fragment reads as the forward conv kernel.
//
// The pixel range is split over grid.z slices; each block atomically
// accumulates its fp32 partial tile into dW (output is tiny — K x RSC —
// so atomic traffic is negligible next to the reduction reads).
//
// Adopted per shape only when it measures faster than MIOpen's igemm
// wrw (same policy as conv_mfma.hip).

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.hip.h"

static inline hipStream_t deepof_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

namespace {

using bf16 = __bf16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(8))) short short8v;

// element-index swizzle within a [row][64] (128-B row) tile — same
// family as conv_mfma.hip (conflict-free for 16-consecutive-row b128
// fragment groups)
__device__ inline int wswz(int row, int col) {
  return col ^ (((row >> 1) & 7) << 3);
}

// BKP = pixels per stage (the MFMA reduction tile)
template <int BKP>
__global__ __launch_bounds__(256)
void conv_wrw_mfma_kernel(const bf16* __restrict__ gy,  // [B,OH,OW,K]
                          const bf16* __restrict__ x,   // [B,IH,IW,C]
                          float* __restrict__ dw,       // [K, RSC] fp32
                          int B, int IH, int IW, int C, int K,
                          int R, int S, int OH, int OW,
                          int stride, int pad,
                          int n_tiles_n, int pix_per_slice, int use_swz) {
  // LDS: double-buffered gyT [64][BKP] + xT [64][BKP] bf16
  __shared__ bf16 lds_all[2 * 2 * 64 * BKP];
#define LDS_GY(buf) (lds_all + (buf) * 64 * BKP)
#define LDS_X(buf) (lds_all + 2 * 64 * BKP + (buf) * 64 * BKP)

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

  // rsc -> (r, s, c): the 64-wide rsc tile stays inside one (r,s) tap
  // when C % 64 == 0 (host guarantees)
  const int rs = tile_n / C;
  const int rr = rs / S, ss = rs % S;
  const int c0 = tile_n % C;

  // ---- staging geometry ----
  // 256 threads transpose a [BKP pix][64 ch] global tile into a
  // [64 ch][BKP pix] LDS image.  Thread handles PIXPAIRS pixel-pairs
  // x one 8-channel chunk: loads 2 x bf16x8, writes 8 x b32
  // (2 pixels packed per 32-bit write, consecutive pix columns).
  // layout: thread t -> chunk8 = t % 8 (8 chunks of 8 channels),
  //                     pair = t / 8 (BKP/2 pairs need BKP/2 threads/chunk)
  constexpr int PAIRS = BKP / 2;             // pixel pairs per tile
  const int st_c8 = (tid & 7) * 8;           // channel chunk base
  const int st_pair = tid >> 3;              // 0 .. 31 (256/8)
  constexpr int PAIR_STEP = 32;              // pairs covered per pass

  auto stage = [&](int pix_base, int buf) {
    for (int pr = st_pair; pr < PAIRS; pr += PAIR_STEP) {
      const int p0 = pix_base + pr * 2;
      const int p1 = p0 + 1;
      // gy tile: rows = k
      bf16x8 a0 = {}, a1 = {};
      if (p0 < pix_end && tile_k + st_c8 < K)
        a0 = *reinterpret_cast<const bf16x8*>(
            gy + (long)p0 * K + tile_k + st_c8);
      if (p1 < pix_end && tile_k + st_c8 < K)
        a1 = *reinterpret_cast<const bf16x8*>(
            gy + (long)p1 * K + tile_k + st_c8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int row = st_c8 + j;                     // k row
        const int col = use_swz ? (wswz(row, pr * 2) & (BKP - 1)) : pr * 2;
        // element-typed stores (an unsigned* write here is a TBAA
        // violation against the bf16x8 fragment reads)
        LDS_GY(buf)[row * BKP + col] = a0[j];
        LDS_GY(buf)[row * BKP + col + 1] = a1[j];
      }
      // x tile: rows = rsc (c within the tap)
      auto src_px = [&](int p, bf16x8& out) {
        if (p >= pix_end) return;
        const int ox = p % OW;
        const int oy = (p / OW) % OH;
        const int bb = p / (OW * OH);
        const int iy = oy * stride + rr - pad;
        const int ix = ox * stride + ss - pad;
        if (iy >= 0 && iy < IH && ix >= 0 && ix < IW)
          out = *reinterpret_cast<const bf16x8*>(
              x + (((long)bb * IH + iy) * IW + ix) * C + c0 + st_c8);
      };
      bf16x8 b0 = {}, b1 = {};
      src_px(p0, b0);
      src_px(p1, b1);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int row = st_c8 + j;
        const int col = use_swz ? (wswz(row, pr * 2) & (BKP - 1)) : pr * 2;
        LDS_X(buf)[row * BKP + col] = b0[j];
        LDS_X(buf)[row * BKP + col + 1] = b1[j];
      }
    }
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
      for (int mi = 0; mi < 2; ++mi) {
        const int row = wm + mi * 16 + (lane & 15);
        const int col = (kk + (lane >> 4) * 8);
        afrag[mi] = *reinterpret_cast<const bf16x8*>(
            &LDS_GY(buf)[row * BKP +
                         (use_swz ? (wswz(row, col) & (BKP - 1)) : col)]);
      }
#pragma unroll
      for (int ni = 0; ni < 2; ++ni) {
        const int row = wn + ni * 16 + (lane & 15);
        const int col = (kk + (lane >> 4) * 8);
        bfrag[ni] = *reinterpret_cast<const bf16x8*>(
            &LDS_X(buf)[row * BKP +
                        (use_swz ? (wswz(row, col) & (BKP - 1)) : col)]);
      }
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
#undef LDS_GY
#undef LDS_X
}

}  // namespace

// gy [B,K,OH,OW] channels_last, x [B,C,IH,IW] channels_last ->
// dW [K,C,R,S] channels_last bf16
at::Tensor conv2d_wrw(at::Tensor gy, at::Tensor x, long R_, long S_,
                      long stride, long pad) {
  TORCH_CHECK(gy.is_cuda() && x.is_cuda());
  TORCH_CHECK(gy.scalar_type() == at::kBFloat16 &&
              x.scalar_type() == at::kBFloat16);
  TORCH_CHECK(gy.is_contiguous(at::MemoryFormat::ChannelsLast) &&
              x.is_contiguous(at::MemoryFormat::ChannelsLast));
  const int B = x.size(0), C = x.size(1), IH = x.size(2), IW = x.size(3);
  const int K = gy.size(1), OH = gy.size(2), OW = gy.size(3);
  const int R = (int)R_, S = (int)S_;
  TORCH_CHECK(C % 64 == 0, "conv2d_wrw needs C % 64 == 0");
  TORCH_CHECK(K % 8 == 0);
  const int RSC = R * S * C;
  const long M = (long)B * OH * OW;

  auto dw_f32 = at::zeros({K, RSC}, x.options().dtype(at::kFloat));
  const int ktiles = (K + 63) / 64;
  const int ntiles = RSC / 64;
  // pick the pixel split so the grid fills the chip (~2 blocks/CU)
  int split = (int)std::max(1L, 512L / ((long)ktiles * ntiles));
  constexpr int BKP = 64;
  const int pix_per_slice =
      (int)(((M + split - 1) / split + BKP - 1) / BKP) * BKP;
  split = (int)((M + pix_per_slice - 1) / pix_per_slice);

  const dim3 grid(ktiles * ntiles, split), block(256);
  static int use_swz = [] {
    const char* e = getenv("DEEPOF_WRW_NOSWZ");
    return e && atoi(e) ? 0 : 1;
  }();
  hipLaunchKernelGGL((conv_wrw_mfma_kernel<BKP>), grid, block, 0,
                     deepof_stream(),
                     reinterpret_cast<const bf16*>(gy.data_ptr()),
                     reinterpret_cast<const bf16*>(x.data_ptr()),
                     dw_f32.data_ptr<float>(), B, IH, IW, C, K, R, S,
                     OH, OW, (int)stride, (int)pad, ntiles,
                     pix_per_slice, use_swz);
  // [K, RSC] fp32 -> [K, C, R, S] channels_last bf16
  auto dw = dw_f32.view({K, R, S, C}).permute({0, 3, 1, 2}).to(at::kBFloat16);
  return dw.contiguous(at::MemoryFormat::ChannelsLast);
}
