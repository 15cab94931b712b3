// FlowNetC correlation cost volume for gfx950.
//
// out[b, d, y, x] = <f1[b,:,y,x], f2[b,:,y+dy,x+dx]> / C for
// (dy, dx) in [-md, md]^2 — absent in the reference TF repo, required
// by BASELINE.json configs[2] (441 channels at md=10).
//
// Strategy: repack NCHW -> NHWC once (channel vectors contiguous), then
// one workgroup per (b, y, x-tile): the tile's f1 vectors are staged in
// LDS (each re-used by all (2md+1)^2 displacements), f2 rows stream
// through L2.  fp32 compute.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.hip.h"

static inline hipStream_t deepof_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

namespace {

// NCHW -> NHWC (fp32)
template <typename T>
__global__ void nchw_to_nhwc(const T* __restrict__ in, float* __restrict__ out,
                             int B, int C, int HW) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long)B * C * HW) return;
  const int c = (idx / HW) % C;
  const int b = idx / ((long)C * HW);
  const int p = idx % HW;
  out[((long)b * HW + p) * C + c] = static_cast<float>(in[idx]);
}

// One block = TILE output pixels of one row; f1 tile staged in LDS.
// blockDim.x = 256; displacements split across threads.
template <int TILE>
__global__ void corr_fwd_kernel(const float* __restrict__ f1,  // NHWC
                                const float* __restrict__ f2,  // NHWC
                                float* __restrict__ out,       // NCHW [B,K2,H,W]
                                int B, int C, int H, int W, int md) {
  const int K = 2 * md + 1;
  const int K2 = K * K;
  extern __shared__ float lds_f1[];  // TILE * C

  const int tiles_x = (W + TILE - 1) / TILE;
  const int b = blockIdx.x / (H * tiles_x);
  const int rem = blockIdx.x % (H * tiles_x);
  const int y = rem / tiles_x;
  const int tx0 = (rem % tiles_x) * TILE;

  // stage f1[b, y, tx0:tx0+TILE, :] into LDS (coalesced over C)
  const long f1_base = ((long)b * H * W + (long)y * W + tx0) * C;
  const int n_stage = TILE * C;
  for (int i = threadIdx.x; i < n_stage; i += blockDim.x) {
    const int px = i / C;
    if (tx0 + px < W) lds_f1[i] = f1[f1_base + i];
  }
  __syncthreads();

  // each thread: one (pixel-in-tile, displacement) pair, strided
  const float inv_c = 1.0f / C;
  for (int work = threadIdx.x; work < TILE * K2; work += blockDim.x) {
    const int px = work / K2;
    const int x = tx0 + px;
    if (x >= W) continue;
    const int d = work % K2;
    const int dy = d / K - md;
    const int dx = d % K - md;
    const int yy = y + dy, xx = x + dx;
    float acc = 0.f;
    if (yy >= 0 && yy < H && xx >= 0 && xx < W) {
      const float* v1 = lds_f1 + px * C;
      const float* v2 = f2 + ((long)b * H * W + (long)yy * W + xx) * C;
      int c = 0;
      for (; c + 4 <= C; c += 4) {
        const float4 a = *reinterpret_cast<const float4*>(v1 + c);
        const float4 bb = *reinterpret_cast<const float4*>(v2 + c);
        acc += a.x * bb.x + a.y * bb.y + a.z * bb.z + a.w * bb.w;
      }
      for (; c < C; ++c) acc += v1[c] * v2[c];
    }
    out[(((long)b * K2 + d) * H + y) * W + x] = acc * inv_c;
  }
}

// grad f1[b,:,y,x] = sum_d gout[b,d,y,x] * f2[b,:,y+dy,x+dx] / C
// grad f2[b,:,y,x] = sum_d gout[b,d,y-dy,x-dx] * f1[b,:,y-dy,x-dx] / C
// One thread per (b, pixel, c-chunk of 4).
template <bool FOR_F2>
__global__ void corr_bwd_kernel(const float* __restrict__ gout,  // NCHW K2
                                const float* __restrict__ other, // NHWC
                                float* __restrict__ grad,        // NHWC out
                                int B, int C, int H, int W, int md) {
  const int K = 2 * md + 1;
  const int K2 = K * K;
  const int c4 = C / 4;  // C is a multiple of 4 for all our encoders
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (idx >= (long)B * H * W * c4) return;
  const int cc = (idx % c4) * 4;
  const long pix = idx / c4;
  const int b = pix / (H * W);
  const int p = pix % (H * W);
  const int y = p / W, x = p % W;

  float4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int d = 0; d < K2; ++d) {
    const int dy = d / K - md;
    const int dx = d % K - md;
    // FOR_F2: source pixel is (y-dy, x-dx); for f1 it's (y, x) itself
    const int sy = FOR_F2 ? y - dy : y;
    const int sx = FOR_F2 ? x - dx : x;
    const int oy = FOR_F2 ? y - dy : y + dy;  // other-tensor pixel
    const int ox = FOR_F2 ? x - dx : x + dx;
    if (sy < 0 || sy >= H || sx < 0 || sx >= W) continue;
    if (FOR_F2) {
      // other = f1 at (sy, sx); gout at (sy, sx)
      const float g = gout[(((long)b * K2 + d) * H + sy) * W + sx];
      if (g != 0.f) {
        const float4 o = *reinterpret_cast<const float4*>(
            other + ((long)b * H * W + (long)sy * W + sx) * C + cc);
        acc.x += g * o.x; acc.y += g * o.y; acc.z += g * o.z; acc.w += g * o.w;
      }
    } else {
      if (oy < 0 || oy >= H || ox < 0 || ox >= W) continue;
      const float g = gout[(((long)b * K2 + d) * H + y) * W + x];
      const float4 o = *reinterpret_cast<const float4*>(
          other + ((long)b * H * W + (long)oy * W + ox) * C + cc);
      acc.x += g * o.x; acc.y += g * o.y; acc.z += g * o.z; acc.w += g * o.w;
    }
  }
  const float inv_c = 1.0f / C;
  float* gp = grad + ((long)b * H * W + p) * C + cc;
  gp[0] = acc.x * inv_c; gp[1] = acc.y * inv_c;
  gp[2] = acc.z * inv_c; gp[3] = acc.w * inv_c;
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

}  // namespace

static inline int iceil2(long a, int b) { return (int)((a + b - 1) / b); }

static at::Tensor to_nhwc_f32(at::Tensor x) {
  const int B = x.size(0), C = x.size(1), HW = x.size(2) * x.size(3);
  auto out = at::empty({B, x.size(2), x.size(3), C},
                       x.options().dtype(at::kFloat));
  const long n = (long)B * C * HW;
  AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf,
      x.scalar_type(), "to_nhwc", [&] {
    hipLaunchKernelGGL(nchw_to_nhwc<scalar_t>, dim3(iceil2(n, 256)), dim3(256),
                       0, deepof_stream(), x.data_ptr<scalar_t>(),
                       out.data_ptr<float>(), B, C, HW);
  });
  return out;
}

at::Tensor correlation_forward(at::Tensor f1, at::Tensor f2, long md) {
  TORCH_CHECK(f1.sizes() == f2.sizes());
  const int B = f1.size(0), C = f1.size(1), H = f1.size(2), W = f1.size(3);
  TORCH_CHECK(C % 4 == 0, "correlation needs C % 4 == 0, got ", C);
  const int K = 2 * (int)md + 1;
  auto f1h = to_nhwc_f32(f1);
  auto f2h = to_nhwc_f32(f2);
  auto out = at::empty({B, K * K, H, W}, f1.options().dtype(at::kFloat));

  constexpr int TILE = 8;
  const int tiles_x = (W + TILE - 1) / TILE;
  const dim3 grid(B * H * tiles_x), block(256);
  const size_t lds = TILE * C * sizeof(float);
  hipLaunchKernelGGL((corr_fwd_kernel<TILE>), grid, block, lds,
                     deepof_stream(), f1h.data_ptr<float>(),
                     f2h.data_ptr<float>(), out.data_ptr<float>(),
                     B, C, H, W, (int)md);
  return out.to(f1.scalar_type());
}

std::vector<at::Tensor> correlation_backward(at::Tensor gout, at::Tensor f1,
                                             at::Tensor f2, long md) {
  const int B = f1.size(0), C = f1.size(1), H = f1.size(2), W = f1.size(3);
  auto goutf = gout.to(at::kFloat).contiguous();
  auto f1h = to_nhwc_f32(f1);
  auto f2h = to_nhwc_f32(f2);
  auto g1h = at::empty_like(f1h);
  auto g2h = at::empty_like(f2h);
  const long n = (long)B * H * W * (C / 4);
  const dim3 grid(iceil2(n, 256)), block(256);
  hipLaunchKernelGGL((corr_bwd_kernel<false>), grid, block, 0,
                     deepof_stream(), goutf.data_ptr<float>(),
                     f2h.data_ptr<float>(), g1h.data_ptr<float>(),
                     B, C, H, W, (int)md);
  hipLaunchKernelGGL((corr_bwd_kernel<true>), grid, block, 0,
                     deepof_stream(), goutf.data_ptr<float>(),
                     f1h.data_ptr<float>(), g2h.data_ptr<float>(),
                     B, C, H, W, (int)md);

  auto back = [&](at::Tensor nhwc, at::ScalarType st) {
    auto out = at::empty({B, C, H, W},
                         f1.options().dtype(st));
    const long nn = (long)B * C * H * W;
    AT_DISPATCH_FLOATING_TYPES_AND2(at::kBFloat16, at::kHalf, st,
        "to_nchw", [&] {
      hipLaunchKernelGGL(nhwc_to_nchw<scalar_t>, dim3(iceil2(nn, 256)),
                         dim3(256), 0, deepof_stream(),
                         nhwc.data_ptr<float>(), out.data_ptr<scalar_t>(),
                         B, C, H * W);
    });
    return out;
  };
  return {back(g1h, f1.scalar_type()), back(g2h, f2.scalar_type())};
}
