// Hand-written MFMA implicit-GEMM convolution for gfx950 (CDNA4).
//
// Forward conv + bias + ELU fused, NHWC bf16 activations, fp32
// accumulate.  GEMM view: O[M=B*OH*OW, N=K] = A[M, RSC] * B[RSC, N];
// the K-loop walks (r, s) filter taps x C-chunks, so every A sub-tile
// load is a contiguous NHWC channel run (coalesced 16-B lane loads)
// and B sub-tiles are contiguous rows of the [K][R][S][C] weight
// (channels_last Conv2d layout).
//
// Tiling: 256 threads = 4 waves (2x2), block tile BM=128 pixels x
// BN=128 channels, each wave a 64x64 sub-tile = 4x4 fragments of
// v_mfma_f32_16x16x32_bf16, BK=64 reduction per stage, double-buffered
// LDS (A 128x64 + B 128x64 bf16 = 32 KiB per stage) with the T2 XOR
// swizzle on both tiles so ds_read_b128 is bank-conflict-free.
//
// Replaces (when faster — the host autotuner decides per shape) the
// MIOpen igemm path for the FlowNetS/VGG encoder stacks
// (/root/reference/flyingChairsWrapFlow.py:31-40) and fuses the ELU
// that PyTorch otherwise runs as a separate elementwise pass.

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

// activation: 0 = none, 1 = ELU, 2 = LeakyReLU(0.1), 3 = ReLU
template <int ACT>
__device__ inline float act_fn(float x) {
  if (ACT == 1) return x > 0.f ? x : expf(x) - 1.f;
  if (ACT == 2) return x > 0.f ? x : 0.1f * x;
  if (ACT == 3) return x > 0.f ? x : 0.f;
  return x;
}

// XOR swizzle on bf16 element index within a [ROW][64] (128-B row)
// tile.  A b128 fragment read group is 16 consecutive rows at one
// 16-B column chunk; its 16-B slot index (addr bits 4-7) is
// (chunk ^ f(row)) | ((row&1)<<3), so f(row) = (row>>1)&7 makes all
// 16 rows land on 16 distinct slots of the 256-B bank row ->
// conflict-free ds_read_b128 ((row&7) alone leaves r / r+8 2-way).
__device__ inline int swz(int row, int col) {
  return col ^ (((row >> 1) & 7) << 3);
}

// ---------------------------------------------------------------------
// fused conv+bias+act forward
//   x:   [B, IH, IW, C]   (NHWC bf16)
//   w:   [K, R, S, C]     (channels_last Conv2d weight, bf16)
//   out: [B, OH, OW, K]   (NHWC bf16)
// Grid: (ceil(M/BM) * ceil(K/BN)) blocks, 256 threads.
// Requires C % CBK == 0 where CBK = min(C, BK).
// ---------------------------------------------------------------------
// STRIDED=true: sub-pixel (transposed-conv) epilogue — the M grid
// enumerates quarter-resolution positions (ty, tx) and each output
// lands at (ty*ostride + off_y, tx*ostride + off_x) of a full-size
// [B, out_cstride, OHf, OWf] buffer at channel offset out_coff.  Four
// parity launches of this variant ARE a stride-2 deconv forward or a
// stride-2 conv backward-data (zero-insertion-free): each parity is a
// stride-1 conv with its own sub-filter/pad (host builds the plan).
// PARITY4=true: all four parities of a sub-pixel transposed conv in ONE
// launch — blockIdx.y selects the parity, whose geometry {packed-weight
// offset, R, S, pad_y, pad_x, off_y, off_x} comes from the 8-int rows
// of `ptab` (device memory, built once per layer plan).  This removes
// the 4-launch + per-launch weight-gather overhead that dominates the
// small decoder layers.
template <int BM, int BN, int BK, int ACT, bool GLDS, bool STRIDED = false,
          bool PARITY4 = false>
__global__ __launch_bounds__(256)
void conv_fwd_mfma_kernel(const bf16* __restrict__ x,
                          const bf16* __restrict__ w,
                          const float* __restrict__ bias,
                          bf16* __restrict__ out,
                          const bf16* __restrict__ zero_page,
                          int B, int IH, int IW, int C,
                          int K, int R, int S, int OH, int OW,
                          int stride, int pad, int n_tiles_n,
                          int pad_x = 0, int ostride = 1, int off_y = 0,
                          int off_x = 0, int OHf = 0, int OWf = 0,
                          int out_cstride = 0, int out_coff = 0,
                          const int* __restrict__ ptab = nullptr) {
  __shared__ bf16 lds_all[2 * (BM + BN) * BK];
  // pointer-array init from addrspace(3) is rejected; index arithmetic
#define LDS_A(buf) (lds_all + (buf) * BM * BK)
#define LDS_B(buf) (lds_all + 2 * BM * BK + (buf) * BN * BK)

  // T1 XCD-aware remap (bijective): consecutive tiles share A-rows /
  // B-panels; keep them on one XCD's L2.  8 XCDs on MI355X.
  int bid = blockIdx.x;
  {
    const int nwg = gridDim.x;
    const int q = nwg / 8, rr = nwg % 8;
    const int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  if constexpr (PARITY4) {
    const int* pp = ptab + blockIdx.y * 8;
    w += pp[0];
    R = pp[1];
    S = pp[2];
    pad = pp[3];
    pad_x = pp[4];
    off_y = pp[5];
    off_x = pp[6];
    OH = (OHf - off_y + ostride - 1) / ostride;  // parity M grid
    OW = (OWf - off_x + ostride - 1) / ostride;
  }
  const int tile_m = bid / n_tiles_n;
  const int tile_n = bid % n_tiles_n;
  const int m0 = tile_m * BM;
  const int n0 = tile_n * BN;
  const int M = B * OH * OW;
  const int padx = STRIDED ? pad_x : pad;      // asymmetric parity pads

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;                    // 4 waves
  constexpr int WAVES_N = BN / 64;             // 2 for BN=128, 1 for BN=64
  constexpr int WAVES_M = 4 / WAVES_N;
  constexpr int M_FRAGS = BM / WAVES_M / 16;   // 16x16 fragments per wave
  constexpr int N_FRAGS = 4;                   // 64 cols per wave
  const int wm = (wid / WAVES_N) * (BM / WAVES_M);
  const int wn = (wid % WAVES_N) * 64;

  const int n_stages = (R * S * C) / BK; // total K stages

  f32x4 acc[M_FRAGS][N_FRAGS] = {};

  // =====================================================================
  // GLDS staging path (requires C % BK == 0): each wave issues 1-KiB
  // global_load_lds_dwordx4 groups (8 rows x 128 B); the T2 swizzle is
  // applied on the per-lane SOURCE address (rule 21 - LDS dest stays
  // lane-linear), OOB taps and padded rows read a 128-B zero page.
  // =====================================================================
  // per-lane fixed geometry
  const int g_row_in_grp = lane >> 3;          // 0..7
  const int g_colb = (lane & 7) * 8;           // element col of this lane
  constexpr int A_GROUPS = BM / 8;             // 1-KiB groups in A tile
  constexpr int B_GROUPS = BN / 8;

  // A pixel coords per group handled by this wave (row fixed per group)
  int ga_iy[A_GROUPS / 4], ga_ix[A_GROUPS / 4];
  long ga_base[A_GROUPS / 4];  // b*IH*IW offset or -1
  if (GLDS) {
#pragma unroll
    for (int gi = 0; gi < A_GROUPS / 4; ++gi) {
      const int g = wid + gi * 4;
      const int row = g * 8 + g_row_in_grp;
      const int m = m0 + row;
      const int mm = m < M ? m : 0;
      const int ox = mm % OW;
      const int oy = (mm / OW) % OH;
      const int bb = mm / (OW * OH);
      ga_iy[gi] = oy * stride - pad;
      ga_ix[gi] = ox * stride - padx;
      ga_base[gi] = (m < M) ? (long)bb * IH * IW : -1;
    }
  }

  auto stage_glds = [&](int stage_idx, int buf) {
    const int rsc0 = stage_idx * BK;
    const int rs = rsc0 / C;       // uniform: C % BK == 0
    const int r = rs / S, ss = rs % S;
    const int c0 = rsc0 % C;
#pragma unroll
    for (int gi = 0; gi < A_GROUPS / 4; ++gi) {
      const int g = wid + gi * 4;
      const int row = g * 8 + g_row_in_grp;
      const int col = g_colb ^ (((row >> 1) & 7) << 3);  // source swizzle
      const int iy = ga_iy[gi] + r;
      const int ix = ga_ix[gi] + ss;
      const bf16* src = zero_page;
      if (ga_base[gi] >= 0 && iy >= 0 && iy < IH && ix >= 0 && ix < IW)
        src = x + (ga_base[gi] + (long)iy * IW + ix) * C + c0 + col;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)
              &LDS_A(buf)[g * 8 * BK],
          16, 0, 0);
    }
#pragma unroll
    for (int gi = 0; gi < B_GROUPS / 4; ++gi) {
      const int g = wid + gi * 4;
      const int row = g * 8 + g_row_in_grp;
      const int col = g_colb ^ (((row >> 1) & 7) << 3);
      const int n = n0 + row;
      const bf16* src = (n < K)
          ? w + (long)n * (R * S * C) + rsc0 + col
          : zero_page;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)
              &LDS_B(buf)[g * 8 * BK],
          16, 0, 0);
    }
  };

  // =====================================================================
  // generic staging path (any C % 8 == 0): plain loads + ds_write
  // =====================================================================
  constexpr int KCHUNKS = BK / 8;
  constexpr int ROWS_PER_PASS = 256 / KCHUNKS;
  const int lrow = tid / KCHUNKS;
  const int lk = (tid % KCHUNKS) * 8;
  int a_b[BM / ROWS_PER_PASS], a_oy[BM / ROWS_PER_PASS],
      a_ox[BM / ROWS_PER_PASS];
  if (!GLDS) {
#pragma unroll
    for (int p = 0; p < BM / ROWS_PER_PASS; ++p) {
      const int m = m0 + lrow + p * ROWS_PER_PASS;
      const int mm = m < M ? m : M - 1;
      a_ox[p] = mm % OW;
      a_oy[p] = (mm / OW) % OH;
      a_b[p] = mm / (OW * OH);
      if (m >= M) a_b[p] = -1;
    }
  }

  auto stage = [&](int stage_idx, int buf) {
#pragma unroll
    for (int p = 0; p < BM / ROWS_PER_PASS; ++p) {
      const int rsc = stage_idx * BK + lk;
      const int c = rsc % C;
      const int rs = rsc / C;
      const int s = rs % S, r = rs / S;
      bf16x8 v = {};
      if (a_b[p] >= 0) {
        const int iy = a_oy[p] * stride + r - pad;
        const int ix = a_ox[p] * stride + s - padx;
        if (iy >= 0 && iy < IH && ix >= 0 && ix < IW) {
          const bf16* src = x + (((long)a_b[p] * IH + iy) * IW + ix) * C + c;
          v = *reinterpret_cast<const bf16x8*>(src);
        }
      }
      const int row = lrow + p * ROWS_PER_PASS;
      *reinterpret_cast<bf16x8*>(&LDS_A(buf)[row * BK + swz(row, lk)]) = v;
    }
#pragma unroll
    for (int p = 0; p < BN / ROWS_PER_PASS; ++p) {
      const int n = n0 + lrow + p * ROWS_PER_PASS;
      const int rsc = stage_idx * BK + lk;
      bf16x8 v = {};
      if (n < K) {
        const bf16* src = w + (long)n * (R * S * C) + rsc;
        v = *reinterpret_cast<const bf16x8*>(src);
      }
      const int row = lrow + p * ROWS_PER_PASS;
      *reinterpret_cast<bf16x8*>(&LDS_B(buf)[row * BK + swz(row, lk)]) = v;
    }
  };

  if (GLDS) stage_glds(0, 0); else stage(0, 0);
  __syncthreads();

  for (int st = 0; st < n_stages; ++st) {
    const int buf = st & 1;
    if (st + 1 < n_stages) {
      if (GLDS) stage_glds(st + 1, buf ^ 1); else stage(st + 1, buf ^ 1);
    }

    // MFMA over this stage: BK reduction = BK/32 mfma k-steps
#pragma unroll
    for (int kk = 0; kk < BK; kk += 32) {
      // load fragments: lane holds 8 contiguous reduction elems
      // A: row = wm + mi*16 + (lane&15), k = kk + (lane>>4)*8
      bf16x8 afrag[M_FRAGS], bfrag[N_FRAGS];
#pragma unroll
      for (int mi = 0; mi < M_FRAGS; ++mi) {
        const int row = wm + mi * 16 + (lane & 15);
        const int col = kk + (lane >> 4) * 8;
        afrag[mi] = *reinterpret_cast<const bf16x8*>(
            &LDS_A(buf)[row * BK + swz(row, col)]);
      }
#pragma unroll
      for (int ni = 0; ni < N_FRAGS; ++ni) {
        const int row = wn + ni * 16 + (lane & 15);
        const int col = kk + (lane >> 4) * 8;
        bfrag[ni] = *reinterpret_cast<const bf16x8*>(
            &LDS_B(buf)[row * BK + swz(row, col)]);
      }
#pragma unroll
      for (int mi = 0; mi < M_FRAGS; ++mi)
#pragma unroll
        for (int ni = 0; ni < N_FRAGS; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              __builtin_bit_cast(short8v, afrag[mi]),
              __builtin_bit_cast(short8v, bfrag[ni]), acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: bias + act, write NHWC ----
  // C/D layout (16x16): col = lane&15, row = (lane>>4)*4 + reg
  if constexpr (!STRIDED) {
#pragma unroll
    for (int mi = 0; mi < M_FRAGS; ++mi) {
#pragma unroll
      for (int ni = 0; ni < N_FRAGS; ++ni) {
        const int n = n0 + wn + ni * 16 + (lane & 15);
        if (n >= K) continue;
        const float bv = bias ? bias[n] : 0.f;
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const int m = m0 + wm + mi * 16 + (lane >> 4) * 4 + reg;
          if (m >= M) continue;
          const float val = act_fn<ACT>(acc[mi][ni][reg] + bv);
          out[(long)m * K + n] = (bf16)val;
        }
      }
    }
  } else {
#pragma unroll
    for (int mi = 0; mi < M_FRAGS; ++mi) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int m = m0 + wm + mi * 16 + (lane >> 4) * 4 + reg;
        if (m >= M) continue;
        const int tx = m % OW;
        const int ty = (m / OW) % OH;
        const int bb = m / (OW * OH);
        const long opix = ((long)bb * OHf + ty * ostride + off_y) * OWf +
                          tx * ostride + off_x;
#pragma unroll
        for (int ni = 0; ni < N_FRAGS; ++ni) {
          const int n = n0 + wn + ni * 16 + (lane & 15);
          if (n >= K) continue;
          const float bv = bias ? bias[n] : 0.f;
          const float val = act_fn<ACT>(acc[mi][ni][reg] + bv);
          out[opix * out_cstride + out_coff + n] = (bf16)val;
        }
      }
    }
  }
#undef LDS_A
#undef LDS_B
}

}  // namespace

// =====================================================================
// Host launcher: x [B,C,H,W] channels_last, w [K,C,R,S] channels_last,
// bias fp32 or undefined, act: 0 none / 1 ELU / 2 LeakyReLU(0.1).
// =====================================================================
at::Tensor conv2d_fwd(at::Tensor x, at::Tensor w, at::Tensor bias,
                      long stride, long pad, long act) {
  TORCH_CHECK(x.is_cuda() && w.is_cuda());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16, "bf16 only");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "x must be channels_last");
  TORCH_CHECK(w.is_contiguous(at::MemoryFormat::ChannelsLast),
              "w must be channels_last");
  const int B = x.size(0), C = x.size(1), IH = x.size(2), IW = x.size(3);
  const int K = w.size(0), R = w.size(2), S = w.size(3);
  TORCH_CHECK(w.size(1) == C);
  TORCH_CHECK(C % 8 == 0, "conv2d_fwd needs C % 8 == 0, got ", C);
  const int OH = (IH + 2 * (int)pad - R) / (int)stride + 1;
  const int OW = (IW + 2 * (int)pad - S) / (int)stride + 1;
  const long M = (long)B * OH * OW;

  constexpr int BM = 128, BK = 64;
  TORCH_CHECK((R * S * C) % BK == 0, "R*S*C must be a multiple of ", BK);

  auto out = at::empty({B, K, OH, OW},
                       x.options().memory_format(at::MemoryFormat::ChannelsLast));
  const float* bptr = nullptr;
  at::Tensor bias_f;
  if (bias.defined() && bias.numel()) {
    bias_f = bias.to(at::kFloat).contiguous();
    bptr = bias_f.data_ptr<float>();
  }

  const bf16* xp = reinterpret_cast<const bf16*>(x.data_ptr());
  const bf16* wp = reinterpret_cast<const bf16*>(w.data_ptr());
  bf16* op = reinterpret_cast<bf16*>(out.data_ptr());

  const int BN = (K >= 128) ? 128 : 64;
  const int n_tiles_n = (K + BN - 1) / BN;
  const long n_blocks = ((M + BM - 1) / BM) * n_tiles_n;
  const dim3 grid((unsigned)n_blocks), block(256);

  // 128-B zero page for OOB taps (glds cannot mask)
  static at::Tensor zero_page;
  if (!zero_page.defined() || zero_page.device() != x.device())
    zero_page = at::zeros({64}, x.options());
  const bf16* zp = reinterpret_cast<const bf16*>(zero_page.data_ptr());
  const bool glds = (C % BK == 0);

#define LAUNCH(BN_, ACT_, GLDS_)                                           \
  hipLaunchKernelGGL((conv_fwd_mfma_kernel<BM, BN_, BK, ACT_, GLDS_>),     \
                     grid, block, 0, deepof_stream(), xp, wp, bptr, op,    \
                     zp, B, IH, IW, C, K, R, S, OH, OW, (int)stride,       \
                     (int)pad, n_tiles_n)
#define LAUNCH_ACT(BN_, GLDS_)                                             \
  do {                                                                     \
    if (act == 1) LAUNCH(BN_, 1, GLDS_);                                   \
    else if (act == 2) LAUNCH(BN_, 2, GLDS_);                              \
    else if (act == 3) LAUNCH(BN_, 3, GLDS_);                              \
    else LAUNCH(BN_, 0, GLDS_);                                            \
  } while (0)
  if (BN == 128) {
    if (glds) LAUNCH_ACT(128, true); else LAUNCH_ACT(128, false);
  } else {
    if (glds) LAUNCH_ACT(64, true); else LAUNCH_ACT(64, false);
  }
#undef LAUNCH_ACT
#undef LAUNCH
  return out;
}

// =====================================================================
// Sub-pixel strided-output conv: one parity of a transposed conv.
//
// Computes a stride-1 conv of x with the (small) parity sub-filter w
// and writes output position (ty, tx) to pixel (ty*ostride + off_y,
// tx*ostride + off_x) of `out` [B, out_cstride, OHf, OWf]
// channels_last, at channel offset out_coff.  Four parity calls
// = a full 2x deconv forward (decoder upconvs,
// /root/reference/flyingChairsWrapFlow.py:65-66) or a stride-2 conv
// backward-data (zero-insertion-free: no dilated gy, no wasted MFMA
// work on zeros).  The host plan (ops/deconv.py) picks per-parity
// sub-filters, pads and offsets.
// =====================================================================
void conv2d_fwd_strided(at::Tensor x, at::Tensor w, at::Tensor bias,
                        at::Tensor out, long pad_y, long pad_x, long act,
                        long ostride, long off_y, long off_x,
                        long out_coff) {
  TORCH_CHECK(x.is_cuda() && w.is_cuda() && out.is_cuda());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 &&
              w.scalar_type() == at::kBFloat16 &&
              out.scalar_type() == at::kBFloat16, "bf16 only");
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(w.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(out.is_contiguous(at::MemoryFormat::ChannelsLast));
  const int B = x.size(0), C = x.size(1), IH = x.size(2), IW = x.size(3);
  const int K = w.size(0), R = w.size(2), S = w.size(3);
  TORCH_CHECK(w.size(1) == C);
  TORCH_CHECK(C % 8 == 0, "needs C % 8 == 0, got ", C);
  constexpr int BM = 128, BK = 64;
  TORCH_CHECK((R * S * C) % BK == 0, "R*S*C must be a multiple of ", BK);
  TORCH_CHECK(out.size(0) == B);
  const int OHf = out.size(2), OWf = out.size(3);
  const int out_cstride = out.size(1);
  TORCH_CHECK(out_coff + K <= out_cstride);
  // the M grid enumerates target pixels ty*ostride+off_y < OHf
  const int MH = (int)((OHf - off_y + ostride - 1) / ostride);
  const int MW = (int)((OWf - off_x + ostride - 1) / ostride);
  const long M = (long)B * MH * MW;

  const float* bptr = nullptr;
  at::Tensor bias_f;
  if (bias.defined() && bias.numel()) {
    bias_f = bias.to(at::kFloat).contiguous();
    bptr = bias_f.data_ptr<float>();
  }
  const bf16* xp = reinterpret_cast<const bf16*>(x.data_ptr());
  const bf16* wp = reinterpret_cast<const bf16*>(w.data_ptr());
  bf16* op = reinterpret_cast<bf16*>(out.data_ptr());

  const int BN = (K >= 128) ? 128 : 64;
  const int n_tiles_n = (K + BN - 1) / BN;
  const long n_blocks = ((M + BM - 1) / BM) * n_tiles_n;
  const dim3 grid((unsigned)n_blocks), block(256);

  static at::Tensor zero_page;
  if (!zero_page.defined() || zero_page.device() != x.device())
    zero_page = at::zeros({64}, x.options());
  const bf16* zp = reinterpret_cast<const bf16*>(zero_page.data_ptr());
  const bool glds = (C % BK == 0);

#define SLAUNCH(BN_, ACT_, GLDS_)                                          \
  hipLaunchKernelGGL(                                                      \
      (conv_fwd_mfma_kernel<BM, BN_, BK, ACT_, GLDS_, true>), grid, block, \
      0, deepof_stream(), xp, wp, bptr, op, zp, B, IH, IW, C, K, R, S,     \
      MH, MW, 1, (int)pad_y, n_tiles_n, (int)pad_x, (int)ostride,          \
      (int)off_y, (int)off_x, OHf, OWf, out_cstride, (int)out_coff)
#define SLAUNCH_ACT(BN_, GLDS_)                                            \
  do {                                                                     \
    if (act == 1) SLAUNCH(BN_, 1, GLDS_);                                  \
    else if (act == 2) SLAUNCH(BN_, 2, GLDS_);                             \
    else if (act == 3) SLAUNCH(BN_, 3, GLDS_);                             \
    else SLAUNCH(BN_, 0, GLDS_);                                           \
  } while (0)
  if (BN == 128) {
    if (glds) SLAUNCH_ACT(128, true); else SLAUNCH_ACT(128, false);
  } else {
    if (glds) SLAUNCH_ACT(64, true); else SLAUNCH_ACT(64, false);
  }
#undef SLAUNCH_ACT
#undef SLAUNCH
}

// =====================================================================
// Sub-pixel parity weight pack: gather the 4 parity sub-filters of a
// transposed conv out of the live weight in ONE launch.
//   src: channels_last [M, N, R, S]  (memory [M][R][S][N]) — for
//        bwd-data this is the conv weight [K, C, R, S] itself; for a
//        deconv the ConvTranspose2d weight [C, K, 4, 4] channels_last.
//   out: flat; parity p at tab[p*12+0] with layout [N][jy][jx][M]
//        (= the conv kernel's [Kout, R', S', Cred] B-operand layout).
//   tab (device int32 [4,12]): {off, nry, nrx, ty0..3, tx0..3, pad}.
// Reads are wave-coalesced along N; each thread writes one 16-B
// bf16x8 chunk of the M-contiguous output row.
// =====================================================================
namespace {

__global__ __launch_bounds__(256)
void subpixel_pack_kernel(const bf16* __restrict__ w, bf16* __restrict__ out,
                          const int* __restrict__ tab, int N, int M,
                          int R, int S) {
  const int* t = tab + blockIdx.y * 12;
  const int off = t[0], nry = t[1], nrx = t[2];
  const int M8 = M >> 3;
  const long total = (long)N * M8 * nry * nrx;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const int n = (int)(i % N);
    long r = i / N;
    const int m8 = (int)(r % M8);
    r /= M8;
    const int jx = (int)(r % nrx);
    const int jy = (int)(r / nrx);
    const int ty = t[3 + jy], tx = t[7 + jx];
    bf16x8 v;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      v[j] = w[(((long)(m8 * 8 + j) * R + ty) * S + tx) * N + n];
    *reinterpret_cast<bf16x8*>(
        out + off + (((long)n * nry + jy) * nrx + jx) * M + m8 * 8) = v;
  }
}

}  // namespace

at::Tensor subpixel_pack(at::Tensor w, at::Tensor tab, long R_, long S_) {
  TORCH_CHECK(w.is_cuda() && w.scalar_type() == at::kBFloat16);
  TORCH_CHECK(w.is_contiguous(at::MemoryFormat::ChannelsLast),
              "pack source must be channels_last [M, N, R, S]");
  TORCH_CHECK(tab.is_cuda() && tab.scalar_type() == at::kInt &&
              tab.numel() == 48);
  const int M = w.size(0), N = w.size(1);
  const int R = (int)R_, S = (int)S_;
  TORCH_CHECK(w.size(2) == R && w.size(3) == S);
  TORCH_CHECK(M % 8 == 0, "pack needs M % 8 == 0");
  auto out = at::empty({(long)M * N * R * S}, w.options());
  const long per_par = (long)N * (M / 8) * R * S;  // upper bound of work
  const int blocks = (int)std::min((per_par + 255) / 256, (long)1024);
  hipLaunchKernelGGL(subpixel_pack_kernel, dim3(blocks, 4), dim3(256), 0,
                     deepof_stream(),
                     reinterpret_cast<const bf16*>(w.data_ptr()),
                     reinterpret_cast<bf16*>(out.data_ptr()),
                     tab.data_ptr<int>(), N, M, R, S);
  return out;
}

// =====================================================================
// All-parity sub-pixel transposed conv in one launch (PARITY4 variant).
//   x: [B, C, IH, IW] channels_last bf16 (the transposed conv's input;
//      gy for backward-data);  wpacked: subpixel_pack output;
//   ptab (device int32 [4,8]): {woff, R, S, pad_y, pad_x, off_y, off_x}.
// =====================================================================
void conv2d_fwd_subpixel4(at::Tensor x, at::Tensor wpacked, at::Tensor bias,
                          at::Tensor out, at::Tensor ptab, long K_, long act,
                          long ostride, long out_coff) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16);
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(out.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(ptab.is_cuda() && ptab.scalar_type() == at::kInt &&
              ptab.numel() == 32);
  const int B = x.size(0), C = x.size(1), IH = x.size(2), IW = x.size(3);
  const int K = (int)K_;
  TORCH_CHECK(C % 64 == 0, "subpixel4 needs C % 64 == 0 (red channels)");
  const int OHf = out.size(2), OWf = out.size(3);
  const int out_cstride = out.size(1);
  TORCH_CHECK(out_coff + K <= out_cstride);
  constexpr int BM = 128;
  const int MH = (OHf + (int)ostride - 1) / (int)ostride;  // max over parities
  const int MW = (OWf + (int)ostride - 1) / (int)ostride;
  const long M = (long)B * MH * MW;

  const float* bptr = nullptr;
  at::Tensor bias_f;
  if (bias.defined() && bias.numel()) {
    bias_f = bias.to(at::kFloat).contiguous();
    bptr = bias_f.data_ptr<float>();
  }
  static at::Tensor zero_page;
  if (!zero_page.defined() || zero_page.device() != x.device())
    zero_page = at::zeros({64}, x.options());

  const bf16* xp = reinterpret_cast<const bf16*>(x.data_ptr());
  const bf16* wp = reinterpret_cast<const bf16*>(wpacked.data_ptr());
  bf16* op = reinterpret_cast<bf16*>(out.data_ptr());
  const bf16* zp = reinterpret_cast<const bf16*>(zero_page.data_ptr());

  const int BN = (K >= 128) ? 128 : 64;
  const int n_tiles_n = (K + BN - 1) / BN;
  const long n_blocks = ((M + BM - 1) / BM) * n_tiles_n;
  const dim3 grid((unsigned)n_blocks, 4), block(256);

#define P4LAUNCH(BN_, ACT_)                                                 \
  hipLaunchKernelGGL(                                                       \
      (conv_fwd_mfma_kernel<BM, BN_, 64, ACT_, true, true, true>), grid,    \
      block, 0, deepof_stream(), xp, wp, bptr, op, zp, B, IH, IW, C, K,     \
      0, 0, MH, MW, 1, 0, n_tiles_n, 0, (int)ostride, 0, 0, OHf, OWf,       \
      out_cstride, (int)out_coff, ptab.data_ptr<int>())
#define P4LAUNCH_ACT(BN_)                                                   \
  do {                                                                      \
    if (act == 1) P4LAUNCH(BN_, 1);                                         \
    else if (act == 2) P4LAUNCH(BN_, 2);                                    \
    else if (act == 3) P4LAUNCH(BN_, 3);                                    \
    else P4LAUNCH(BN_, 0);                                                  \
  } while (0)
  if (BN == 128) P4LAUNCH_ACT(128); else P4LAUNCH_ACT(64);
#undef P4LAUNCH_ACT
#undef P4LAUNCH
}

// =====================================================================
// Deep-pipelined 256x256 conv kernel (8 waves, 4 phases per K-tile).
//
// The 128x128 2-barrier structure above tops out ~700 TF: its
// stage -> vmcnt(0) -> barrier path serializes (cdna guide §5).  This
// kernel applies the guide's 256² deep-pipeline recipe to the conv:
//  - K-tile (BK=64) split into two 32-wide k-halves; A and B halves
//    (256x32 bf16 = 16 KiB each) are staged by global_load_lds one
//    half per phase, FOUR phases ahead of consumption,
//  - counted s_waitcnt vmcnt(4) (never 0 mid-loop) + raw s_barrier:
//    prefetched halves stay in flight across barriers,
//  - s_setprio(1) around each 16-MFMA cluster (phase role split),
//  - LDS swizzle: phys = logical ^ (((row>>2)&3)<<4) bytes -> the 16
//    rows of a b128 fragment-read group cover 16 distinct 16-B slots
//    (conflict-free); applied on the glds SOURCE address and the read.
// LDS: 2 dbuf x 2 khalf x (A 16K + B 16K) = 128 KiB -> 1 block/CU.
// =====================================================================
namespace {

template <int SWZV>
__device__ inline int swz256f(int row) {
  if (SWZV == 1) return (row >> 1) & 3;
  if (SWZV == 2) return ((row >> 1) & 3) ^ ((row >> 3) & 3);
  return (row >> 2) & 3;
}

template <int ACT, int SWZV = 0>
__global__ __launch_bounds__(512)
void conv_fwd_mfma256_kernel(const bf16* __restrict__ x,
                             const bf16* __restrict__ w,
                             const float* __restrict__ bias,
                             bf16* __restrict__ out,
                             const bf16* __restrict__ zero_page,
                             int B, int IH, int IW, int C,
                             int K, int R, int S, int OH, int OW,
                             int stride, int pad, int n_tiles_n) {
  constexpr int BM = 256, BN = 256, BK = 64;
  // one __shared__ object (hipcc trap: a second one forces vmcnt(0)
  // before every ds_read of a glds pipeline)
  __shared__ bf16 lds_all[2 * 2 * (256 * 32) * 2];  // dbuf x khalf x (A+B)
#define LDS256_A(buf, kh) (lds_all + ((buf) * 2 + (kh)) * (256 * 32))
#define LDS256_B(buf, kh) (lds_all + 4 * (256 * 32) + ((buf) * 2 + (kh)) * (256 * 32))

  int bid = blockIdx.x;
  {
    const int nwg = gridDim.x;
    const int q = nwg / 8, rr = nwg % 8;
    const int xcd = bid % 8, idx = bid / 8;
    bid = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int tile_m = bid / n_tiles_n;
  const int tile_n = bid % n_tiles_n;
  const int m0 = tile_m * BM;
  const int n0 = tile_n * BN;
  const int M = B * OH * OW;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;              // 8 waves, 2M x 4N
  const int wm = (wid >> 2) * 128;       // wave row offset (0 / 128)
  const int wn = (wid & 3) * 64;         // wave col offset

  // ---- staging geometry: 1-KiB glds covers 16 rows x 64 B ----
  const int st_row_in_grp = lane >> 2;       // 0..15
  const int st_colb = (lane & 3) * 8;        // element col (of 32)
  // per-wave groups: {wid, wid+8} of 16 rows each
  int sa_iy[2], sa_ix[2];
  long sa_base[2];
  int sb_n[2];
#pragma unroll
  for (int gi = 0; gi < 2; ++gi) {
    const int g = wid + gi * 8;
    const int row = g * 16 + st_row_in_grp;
    const int m = m0 + row;
    const int mm = m < M ? m : 0;
    const int ox = mm % OW;
    const int oy = (mm / OW) % OH;
    const int bb = mm / (OW * OH);
    sa_iy[gi] = oy * stride - pad;
    sa_ix[gi] = ox * stride - pad;
    sa_base[gi] = (m < M) ? (long)bb * IH * IW : -1;
    sb_n[gi] = n0 + row;
  }
  // source swizzle col (elements) for this lane within a 32-col half
  auto swz_col = [&](int row) {
    return st_colb ^ (swz256f<SWZV>(row) << 3);
  };

  const int n_stages = (R * S * C) / BK;

  // stage one half-tile (A or B) of K-tile `t` into dbuf slot t&1
  auto stage_half = [&](int t, int kh, bool is_a) {
    const int rsc0 = t * BK + kh * 32;
    const int rs = rsc0 / C;
    const int r = rs / S, ss = rs % S;
    const int c0 = rsc0 % C;
    const int buf = t & 1;
#pragma unroll
    for (int gi = 0; gi < 2; ++gi) {
      const int g = wid + gi * 8;
      const int row = g * 16 + st_row_in_grp;
      const int col = swz_col(row);
      const bf16* src = zero_page;
      if (is_a) {
        const int iy = sa_iy[gi] + r;
        const int ix = sa_ix[gi] + ss;
        if (sa_base[gi] >= 0 && iy >= 0 && iy < IH && ix >= 0 && ix < IW)
          src = x + (sa_base[gi] + (long)iy * IW + ix) * C + c0 + col;
      } else {
        if (sb_n[gi] < K)
          src = w + (long)sb_n[gi] * (R * S * C) + rsc0 + col;
      }
      bf16* dst = (is_a ? LDS256_A(buf, kh) : LDS256_B(buf, kh)) + g * 16 * 32;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)src,
          (__attribute__((address_space(3))) unsigned int*)dst, 16, 0, 0);
    }
  };

  f32x4 acc[8][4] = {};

  // prologue: all four halves of tile 0 (order matches the loop's)
  stage_half(0, 0, true);   // A_k0
  stage_half(0, 0, false);  // B_k0
  stage_half(0, 1, true);   // A_k1
  stage_half(0, 1, false);  // B_k1

  const int frag_col = (lane >> 4) * 8;  // k offset within the 32-col half

  bf16x8 bfrag[4];
  for (int t = 0; t < n_stages; ++t) {
    const int buf = t & 1;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int kh = p >> 1;   // k-half consumed this phase
      const int mq = p & 1;    // m-frag quadrant (frags 4mq..4mq+3)

      if (p == 0 || p == 2) {
        // own glds for this k-half landed; leaves 4 (2 half-tiles) in
        // flight mid-loop, drains only on the last tile
        if (t + 1 < n_stages)
          asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        __builtin_amdgcn_s_barrier();
      }

      // ds-read this phase's fragments (plain loads: hipcc emits the
      // fine-grained lgkmcnt waits before the MFMAs itself).
      // B fragments are shared by the two m-quadrant phases of a
      // k-half: read only when the quadrant is 0.
      bf16x8 afrag[4];
      const bf16* a_base = LDS256_A(buf, kh);
      const bf16* b_base = LDS256_B(buf, kh);
#pragma unroll
      for (int mf = 0; mf < 4; ++mf) {
        const int row = wm + (mq * 4 + mf) * 16 + (lane & 15);
        const int col = frag_col ^ (swz256f<SWZV>(row) << 3);
        afrag[mf] = *reinterpret_cast<const bf16x8*>(a_base + row * 32 + col);
      }
      if (mq == 0) {
#pragma unroll
        for (int nf = 0; nf < 4; ++nf) {
          const int row = wn + nf * 16 + (lane & 15);
          const int col = frag_col ^ (swz256f<SWZV>(row) << 3);
          bfrag[nf] = *reinterpret_cast<const bf16x8*>(
              b_base + row * 32 + col);
        }
      }

      // issue next tile's half for THIS phase slot (4 phases ahead)
      if (t + 1 < n_stages) {
        if (p == 0) stage_half(t + 1, 0, true);
        else if (p == 1) stage_half(t + 1, 0, false);
        else if (p == 2) stage_half(t + 1, 1, true);
        else stage_half(t + 1, 1, false);
      }

      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mf = 0; mf < 4; ++mf)
#pragma unroll
        for (int nf = 0; nf < 4; ++nf)
          acc[mq * 4 + mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              __builtin_bit_cast(short8v, afrag[mf]),
              __builtin_bit_cast(short8v, bfrag[nf]),
              acc[mq * 4 + mf][nf], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
  }

  // ---- epilogue ----
#pragma unroll
  for (int mf = 0; mf < 8; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const int n = n0 + wn + nf * 16 + (lane & 15);
      if (n >= K) continue;
      const float bv = bias ? bias[n] : 0.f;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int m = m0 + wm + mf * 16 + (lane >> 4) * 4 + reg;
        if (m >= M) continue;
        const float val = act_fn<ACT>(acc[mf][nf][reg] + bv);
        out[(long)m * K + n] = (bf16)val;
      }
    }
  }
#undef LDS256_A
#undef LDS256_B
}

}  // namespace

at::Tensor conv2d_fwd256(at::Tensor x, at::Tensor w, at::Tensor bias,
                         long stride, long pad, long act) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16);
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(w.is_contiguous(at::MemoryFormat::ChannelsLast));
  const int B = x.size(0), C = x.size(1), IH = x.size(2), IW = x.size(3);
  const int K = w.size(0), R = w.size(2), S = w.size(3);
  TORCH_CHECK(C % 32 == 0, "conv2d_fwd256 needs C % 32 == 0");
  TORCH_CHECK((R * S * C) % 64 == 0);
  const int OH = (IH + 2 * (int)pad - R) / (int)stride + 1;
  const int OW = (IW + 2 * (int)pad - S) / (int)stride + 1;
  const long M = (long)B * OH * OW;

  auto out = at::empty({B, K, OH, OW},
                       x.options().memory_format(at::MemoryFormat::ChannelsLast));
  const float* bptr = nullptr;
  at::Tensor bias_f;
  if (bias.defined() && bias.numel()) {
    bias_f = bias.to(at::kFloat).contiguous();
    bptr = bias_f.data_ptr<float>();
  }
  static at::Tensor zero_page;
  if (!zero_page.defined() || zero_page.device() != x.device())
    zero_page = at::zeros({64}, x.options());

  const bf16* xp = reinterpret_cast<const bf16*>(x.data_ptr());
  const bf16* wp = reinterpret_cast<const bf16*>(w.data_ptr());
  bf16* op = reinterpret_cast<bf16*>(out.data_ptr());
  const bf16* zp = reinterpret_cast<const bf16*>(zero_page.data_ptr());

  const int n_tiles_n = (K + 255) / 256;
  const long n_blocks = ((M + 255) / 256) * n_tiles_n;
  const dim3 grid((unsigned)n_blocks), block(512);
  // measured: SWZ variant 1 ((row>>1)&3) is fastest on gfx950
  // (profiles/r01_conv_mfma_pmc.md sweep)
  static int swzv = [] {
    const char* e = getenv("DEEPOF_CONV256_SWZ");
    return e ? atoi(e) : 1;
  }();
#define L256(ACT_, SWZ_) \
  hipLaunchKernelGGL((conv_fwd_mfma256_kernel<ACT_, SWZ_>), grid, block,  \
                     0, deepof_stream(), xp, wp, bptr, op, zp, B, IH, IW, \
                     C, K, R, S, OH, OW, (int)stride, (int)pad, n_tiles_n)
#define L256A(SWZ_)                              \
  do {                                           \
    if (act == 1) L256(1, SWZ_);                 \
    else if (act == 2) L256(2, SWZ_);            \
    else if (act == 3) L256(3, SWZ_);            \
    else L256(0, SWZ_);                          \
  } while (0)
  if (swzv == 1) L256A(1);
  else if (swzv == 2) L256A(2);
  else L256A(0);
#undef L256A
#undef L256
  return out;
}
