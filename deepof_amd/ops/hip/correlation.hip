// FlowNetC correlation cost volume for gfx950 (CDNA4).
//
// out[b, d, y, x] = <f1[b,:,y,x], f2[b,:,y+dy,x+dx]> / C for
// (dy, dx) in [-md, md]^2 — absent in the reference TF repo, required
// by BASELINE.json configs[2] (441 channels at md=10).
//
// v2 design (the v1 thread-per-(pixel,displacement) kernel was
// L2-bound at ~4.6 TF): one block owns a TILE-pixel row segment.
//  fwd: each thread holds ITS pixel's f1 vector in registers (bf16,
//       fp32 accumulation); for each dy the f2 row window
//       (TILE+2md pixels) is staged once in LDS and every (px, dx)
//       dot reads only the 0.5 KiB f2 vector from LDS.
//  bwd: same staging pattern for the two gather forms
//       (d f1 = sum_d g[d,p] f2[p+d];  d f2 = sum_d g[d,p-d] f1[p-d]).
// Channel vectors are contiguous (NHWC repack, bf16).

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.hip.h"

static inline hipStream_t deepof_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

namespace {

using bf16 = __bf16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

// NCHW (any float type) -> NHWC bf16
template <typename T>
__global__ void nchw_to_nhwc_bf16(const T* __restrict__ in,
                                  bf16* __restrict__ out,
                                  int B, int C, int HW) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long)B * C * HW) return;
  const int c = (idx / HW) % C;
  const int b = idx / ((long)C * HW);
  const int p = idx % HW;
  out[((long)b * HW + p) * C + c] = (bf16)(float)in[idx];
}

// NHWC fp32 -> NCHW T
template <typename T>
__global__ void nhwc_to_nchw(const float* __restrict__ in, T* __restrict__ out,
                             int B, int C, int HW) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long)B * C * HW) return;
  const int c = (idx / HW) % C;
  const int b = idx / ((long)C * HW);
  const int p = idx % HW;
  out[idx] = static_cast<T>(in[((long)b * HW + p) * C + c]);
}

__device__ inline float dot8(bf16x8 a, bf16x8 b) {
  float s = 0.f;
#pragma unroll
  for (int i = 0; i < 8; ++i) s += (float)a[i] * (float)b[i];
  return s;
}

// ---------------------------------------------------------------------
// forward: block = (b, y, TILE pixels); thread t = (px = t/K, dx = t%K)
// f1 pixel vector lives in registers (C <= CMAX bf16 = C/8 bf16x8 regs)
// ---------------------------------------------------------------------
template <int CMAX>
__global__ __launch_bounds__(256)
void corr_fwd_v2(const bf16* __restrict__ f1,  // NHWC
                 const bf16* __restrict__ f2,  // NHWC
                 float* __restrict__ out,      // NCHW [B, K2, H, W]
                 int B, int C, int H, int W, int md, int tile, int tiles_x) {
  extern __shared__ bf16 lds_f2[];  // (tile + 2*md) * C
  const int K = 2 * md + 1;
  const int K2 = K * K;
  const int win = tile + 2 * md;

  const int b = blockIdx.x / (H * tiles_x);
  const int rem = blockIdx.x % (H * tiles_x);
  const int y = rem / tiles_x;
  const int x0 = (rem % tiles_x) * tile;

  const int t = threadIdx.x;
  const int px = t / K;     // pixel within tile
  const int dx = t % K;     // displacement column (0..K-1)
  const bool active = px < tile && (x0 + px) < W;

  // load this thread's f1 vector into registers
  bf16x8 f1r[CMAX / 8];
  if (active) {
    const bf16* v = f1 + ((long)b * H * W + (long)y * W + x0 + px) * C;
    for (int c8 = 0; c8 < C / 8; ++c8)
      f1r[c8] = *reinterpret_cast<const bf16x8*>(v + c8 * 8);
  }

  for (int dy = -md; dy <= md; ++dy) {
    // stage f2[y+dy, x0-md .. x0-md+win) into LDS (zeros off-image)
    __syncthreads();
    const int yy = y + dy;
    const int n_half = win * C / 8;  // bf16x8 chunks
    for (int i = threadIdx.x; i < n_half; i += blockDim.x) {
      const int p = i / (C / 8);
      const int c8 = i % (C / 8);
      const int xx = x0 - md + p;
      bf16x8 v = {};
      if (yy >= 0 && yy < H && xx >= 0 && xx < W)
        v = *reinterpret_cast<const bf16x8*>(
            f2 + ((long)b * H * W + (long)yy * W + xx) * C + c8 * 8);
      *reinterpret_cast<bf16x8*>(lds_f2 + (long)p * C + c8 * 8) = v;
    }
    __syncthreads();
    if (!active) continue;

    const bf16* v2 = lds_f2 + (long)(px + dx) * C;
    float acc = 0.f;
    for (int c8 = 0; c8 < C / 8; ++c8)
      acc += dot8(f1r[c8],
                  *reinterpret_cast<const bf16x8*>(v2 + c8 * 8));
    const int d = (dy + md) * K + dx;
    out[(((long)b * K2 + d) * H + y) * W + x0 + px] = acc / C;
  }
}

// ---------------------------------------------------------------------
// backward wrt f1: block = (b, y, TB pixels); thread t = (px, c-chunk)
// grad_f1[p, c] = sum_d g[d, p] * f2[p+d, c] / C
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256)
void corr_bwd_f1_v2(const float* __restrict__ g,   // NCHW [B,K2,H,W]
                    const bf16* __restrict__ f2,   // NHWC
                    float* __restrict__ gf1,       // NHWC fp32
                    int B, int C, int H, int W, int md, int tile,
                    int tiles_x) {
  extern __shared__ bf16 lds[];  // f2 window (tile+2md)*C bf16,
                                 // then g tile K*tile floats
  const int K = 2 * md + 1;
  const int win = tile + 2 * md;
  float* g_lds = reinterpret_cast<float*>(lds + (long)win * C);

  const int b = blockIdx.x / (H * tiles_x);
  const int rem = blockIdx.x % (H * tiles_x);
  const int y = rem / tiles_x;
  const int x0 = (rem % tiles_x) * tile;

  const int chunks = C / 8;
  const int t = threadIdx.x;
  const int px = t / chunks;
  const int c8 = t % chunks;
  const bool active = px < tile && (x0 + px) < W;

  float acc[8] = {};
  for (int dy = -md; dy <= md; ++dy) {
    const int yy = y + dy;
    __syncthreads();
    // stage f2 row window
    const int n_half = win * chunks;
    for (int i = threadIdx.x; i < n_half; i += blockDim.x) {
      const int p = i / chunks;
      const int cc = i % chunks;
      const int xx = x0 - md + p;
      bf16x8 v = {};
      if (yy >= 0 && yy < H && xx >= 0 && xx < W)
        v = *reinterpret_cast<const bf16x8*>(
            f2 + ((long)b * H * W + (long)yy * W + xx) * C + cc * 8);
      *reinterpret_cast<bf16x8*>(lds + (long)p * C + cc * 8) = v;
    }
    // stage g[d = (dy, :), y, x0 .. x0+tile)
    for (int i = threadIdx.x; i < K * tile; i += blockDim.x) {
      const int dxx = i / tile;
      const int p = i % tile;
      const int d = (dy + md) * K + dxx;
      float gv = 0.f;
      if (x0 + p < W)
        gv = g[(((long)b * K * K + d) * H + y) * W + x0 + p];
      g_lds[i] = gv;
    }
    __syncthreads();
    if (!active) continue;

    for (int dxx = 0; dxx < K; ++dxx) {
      const float gv = g_lds[dxx * tile + px];
      if (gv == 0.f) continue;
      const bf16x8 v = *reinterpret_cast<const bf16x8*>(
          lds + (long)(px + dxx) * C + c8 * 8);
#pragma unroll
      for (int i = 0; i < 8; ++i) acc[i] += gv * (float)v[i];
    }
  }
  if (active) {
    float* o = gf1 + ((long)b * H * W + (long)y * W + x0 + px) * C + c8 * 8;
#pragma unroll
    for (int i = 0; i < 8; ++i) o[i] = acc[i] / C;
  }
}

// ---------------------------------------------------------------------
// backward wrt f2: grad_f2[p, c] = sum_d g[d, p-d] * f1[p-d, c] / C
// same structure with the f1 row window at y-dy and g at (y-dy, x-dx).
// ---------------------------------------------------------------------
__global__ __launch_bounds__(256)
void corr_bwd_f2_v2(const float* __restrict__ g,
                    const bf16* __restrict__ f1,
                    float* __restrict__ gf2,
                    int B, int C, int H, int W, int md, int tile,
                    int tiles_x) {
  extern __shared__ bf16 lds[];  // f1 window (tile+2md)*C bf16,
                                 // then g tile K*(tile+2md) floats
  const int K = 2 * md + 1;
  const int win = tile + 2 * md;
  float* g_lds = reinterpret_cast<float*>(lds + (long)win * C);

  const int b = blockIdx.x / (H * tiles_x);
  const int rem = blockIdx.x % (H * tiles_x);
  const int y = rem / tiles_x;
  const int x0 = (rem % tiles_x) * tile;

  const int chunks = C / 8;
  const int t = threadIdx.x;
  const int px = t / chunks;
  const int c8 = t % chunks;
  const bool active = px < tile && (x0 + px) < W;

  float acc[8] = {};
  for (int dy = -md; dy <= md; ++dy) {
    const int yy = y - dy;  // source row
    __syncthreads();
    const int n_half = win * chunks;
    for (int i = threadIdx.x; i < n_half; i += blockDim.x) {
      const int p = i / chunks;
      const int cc = i % chunks;
      const int xx = x0 - md + p;
      bf16x8 v = {};
      if (yy >= 0 && yy < H && xx >= 0 && xx < W)
        v = *reinterpret_cast<const bf16x8*>(
            f1 + ((long)b * H * W + (long)yy * W + xx) * C + cc * 8);
      *reinterpret_cast<bf16x8*>(lds + (long)p * C + cc * 8) = v;
    }
    // stage g[(dy,dx), y-dy, x0-md .. x0-md+win) for all dx
    for (int i = threadIdx.x; i < K * win; i += blockDim.x) {
      const int dxx = i / win;
      const int p = i % win;
      const int xx = x0 - md + p;
      const int d = (dy + md) * K + dxx;
      float gv = 0.f;
      if (yy >= 0 && yy < H && xx >= 0 && xx < W)
        gv = g[(((long)b * K * K + d) * H + yy) * W + xx];
      g_lds[i] = gv;
    }
    __syncthreads();
    if (!active) continue;

    for (int dxx = 0; dxx < K; ++dxx) {
      // source pixel q = (y-dy, x-(dxx-md)); local window index
      const int ql = px + md - (dxx - md);
      if (ql < 0 || ql >= win) continue;
      const float gv = g_lds[dxx * win + ql];
      if (gv == 0.f) continue;
      const bf16x8 v = *reinterpret_cast<const bf16x8*>(
          lds + (long)ql * C + c8 * 8);
#pragma unroll
      for (int i = 0; i < 8; ++i) acc[i] += gv * (float)v[i];
    }
  }
  if (active) {
    float* o = gf2 + ((long)b * H * W + (long)y * W + x0 + px) * C + c8 * 8;
#pragma unroll
    for (int i = 0; i < 8; ++i) o[i] = acc[i] / C;
  }
}

}  // namespace

static inline int iceil2(long a, int b) { return (int)((a + b - 1) / b); }

static at::Tensor to_nhwc_b16(at::Tensor x) {
  const int B = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  auto out = at::empty({B, x.size(2), x.size(3), C},
                       x.options().dtype(at::kBFloat16));
  const long n = (long)B * C * HW;
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf,
      x.scalar_type(), "to_nhwc16", [&] {
    hipLaunchKernelGGL(nchw_to_nhwc_bf16<scalar_t>, dim3(iceil2(n, 256)),
                       dim3(256), 0, deepof_stream(),
                       x.data_ptr<scalar_t>(),
                       reinterpret_cast<bf16*>(out.data_ptr()), B, C, HW);
  });
  return out;
}

static at::Tensor from_nhwc_f32(at::Tensor nhwc, at::ScalarType st,
                                int B, int C, int H, int W) {
  auto out = at::empty({B, C, H, W},
                       nhwc.options().dtype(st));
  const long nn = (long)B * C * H * W;
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, st,
      "to_nchw", [&] {
    hipLaunchKernelGGL(nhwc_to_nchw<scalar_t>, dim3(iceil2(nn, 256)),
                       dim3(256), 0, deepof_stream(),
                       nhwc.data_ptr<float>(), out.data_ptr<scalar_t>(),
                       B, C, H * W);
  });
  return out;
}

at::Tensor correlation_forward(at::Tensor f1, at::Tensor f2, long md_) {
  TORCH_CHECK(f1.sizes() == f2.sizes());
  const int B = f1.size(0), C = f1.size(1), H = f1.size(2), W = f1.size(3);
  TORCH_CHECK(C % 8 == 0 && C <= 1024, "correlation needs C % 8 == 0");
  const int md = (int)md_;
  const int K = 2 * md + 1;
  auto f1h = to_nhwc_b16(f1);
  auto f2h = to_nhwc_b16(f2);
  auto out = at::empty({B, K * K, H, W}, f1.options().dtype(at::kFloat));

  const int tile = std::max(1, std::min(256 / K, W));
  const int tiles_x = (W + tile - 1) / tile;
  const dim3 grid((unsigned)((long)B * H * tiles_x)), block(256);
  const size_t lds = (size_t)(tile + 2 * md) * C * sizeof(bf16);
  TORCH_CHECK(lds <= 160 * 1024, "correlation LDS overflow");

#define CORR_FWD(CMAX_)                                                  \
  hipLaunchKernelGGL((corr_fwd_v2<CMAX_>), grid, block, lds,             \
                     deepof_stream(),                                    \
                     reinterpret_cast<const bf16*>(f1h.data_ptr()),      \
                     reinterpret_cast<const bf16*>(f2h.data_ptr()),      \
                     out.data_ptr<float>(), B, C, H, W, md, tile,        \
                     tiles_x)
  if (C <= 64) CORR_FWD(64);
  else if (C <= 128) CORR_FWD(128);
  else if (C <= 256) CORR_FWD(256);
  else CORR_FWD(1024);
#undef CORR_FWD
  return out.to(f1.scalar_type());
}

std::vector<at::Tensor> correlation_backward(at::Tensor gout, at::Tensor f1,
                                             at::Tensor f2, long md_) {
  const int B = f1.size(0), C = f1.size(1), H = f1.size(2), W = f1.size(3);
  const int md = (int)md_;
  const int K = 2 * md + 1;
  auto goutf = gout.to(at::kFloat).contiguous();
  auto f1h = to_nhwc_b16(f1);
  auto f2h = to_nhwc_b16(f2);
  auto g1h = at::empty({B, H, W, C}, f1.options().dtype(at::kFloat));
  auto g2h = at::empty({B, H, W, C}, f1.options().dtype(at::kFloat));

  const int chunks = C / 8;
  const int tile = std::max(1, std::min(256 / chunks, W));
  const int tiles_x = (W + tile - 1) / tile;
  const int win = tile + 2 * md;
  const dim3 grid((unsigned)((long)B * H * tiles_x)), block(256);
  const size_t lds1 = (size_t)win * C * sizeof(bf16) +
                      (size_t)K * tile * sizeof(float);
  const size_t lds2 = (size_t)win * C * sizeof(bf16) +
                      (size_t)K * win * sizeof(float);
  TORCH_CHECK(std::max(lds1, lds2) <= 160 * 1024, "corr bwd LDS overflow");

  hipLaunchKernelGGL(corr_bwd_f1_v2, grid, block, lds1, deepof_stream(),
                     goutf.data_ptr<float>(),
                     reinterpret_cast<const bf16*>(f2h.data_ptr()),
                     g1h.data_ptr<float>(), B, C, H, W, md, tile, tiles_x);
  hipLaunchKernelGGL(corr_bwd_f2_v2, grid, block, lds2, deepof_stream(),
                     goutf.data_ptr<float>(),
                     reinterpret_cast<const bf16*>(f1h.data_ptr()),
                     g2h.data_ptr<float>(), B, C, H, W, md, tile, tiles_x);

  return {from_nhwc_f32(g1h, f1.scalar_type(), B, C, H, W),
          from_nhwc_f32(g2h, f2.scalar_type(), B, C, H, W)};
}
