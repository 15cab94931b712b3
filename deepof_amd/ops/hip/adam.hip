// Fused multi-tensor Adam for gfx950.
//
// One kernel launch updates every parameter tensor of the model
// (the reference runs tf.train.AdamOptimizer's per-variable ops,
// /root/reference/flyingChairsTrain.py:124).  Chunks of all tensors are
// packed into a device-side table; the grid strides over chunks.
// Params/grads/moments are fp32 (master weights).

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>

#include "common.hip.h"

static inline hipStream_t deepof_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

namespace {

constexpr int CHUNK = 1 << 21;  // 2M elements per chunk
constexpr int BLOCK = 256;

struct ChunkInfo {
  float* p;
  float* g;
  float* m;
  float* v;
  int n;
};

// hyper: optional device [lr, bias1, bias2] so a captured hipGraph can
// replay with per-step bias correction (host updates the pinned source).
__global__ void fused_adam_kernel(const ChunkInfo* __restrict__ chunks,
                                  int n_chunks, float lr, float beta1,
                                  float beta2, float eps, float wd,
                                  float bias1, float bias2,
                                  const float* __restrict__ hyper) {
  if (hyper) { lr = hyper[0]; bias1 = hyper[1]; bias2 = hyper[2]; }
  const float step_size = lr / bias1;
  const float inv_sqrt_bias2 = rsqrtf(bias2);
  for (int ci = blockIdx.y; ci < n_chunks; ci += gridDim.y) {
    const ChunkInfo ck = chunks[ci];
    const int base = (blockIdx.x * blockDim.x + threadIdx.x) * 4;
    const int stride = gridDim.x * blockDim.x * 4;
    for (int i = base; i < ck.n; i += stride) {
      // 16B vector path when the tail allows it
      if (i + 4 <= ck.n) {
        float4 p = *reinterpret_cast<float4*>(ck.p + i);
        const float4 gr = *reinterpret_cast<const float4*>(ck.g + i);
        float4 m = *reinterpret_cast<float4*>(ck.m + i);
        float4 v = *reinterpret_cast<float4*>(ck.v + i);
        float pv[4] = {p.x, p.y, p.z, p.w};
        float gv[4] = {gr.x, gr.y, gr.z, gr.w};
        float mv[4] = {m.x, m.y, m.z, m.w};
        float vv[4] = {v.x, v.y, v.z, v.w};
#pragma unroll
        for (int k = 0; k < 4; ++k) {
          float g = gv[k] + wd * pv[k];
          mv[k] = beta1 * mv[k] + (1.0f - beta1) * g;
          vv[k] = beta2 * vv[k] + (1.0f - beta2) * g * g;
          const float denom = sqrtf(vv[k]) * inv_sqrt_bias2 + eps;
          pv[k] -= step_size * mv[k] / denom;
        }
        *reinterpret_cast<float4*>(ck.p + i) =
            make_float4(pv[0], pv[1], pv[2], pv[3]);
        *reinterpret_cast<float4*>(ck.m + i) =
            make_float4(mv[0], mv[1], mv[2], mv[3]);
        *reinterpret_cast<float4*>(ck.v + i) =
            make_float4(vv[0], vv[1], vv[2], vv[3]);
      } else {
        for (int k = i; k < ck.n; ++k) {
          float g = ck.g[k] + wd * ck.p[k];
          ck.m[k] = beta1 * ck.m[k] + (1.0f - beta1) * g;
          ck.v[k] = beta2 * ck.v[k] + (1.0f - beta2) * g * g;
          const float denom = sqrtf(ck.v[k]) * inv_sqrt_bias2 + eps;
          ck.p[k] -= step_size * ck.m[k] / denom;
        }
      }
    }
  }
}

}  // namespace

// Build the device-side chunk table once; pointers stay valid across
// steps (params/moments update in place), so the step itself is a pure
// device launch — hipGraph-capturable.
at::Tensor build_adam_table(std::vector<at::Tensor> params,
                            std::vector<at::Tensor> grads,
                            std::vector<at::Tensor> exp_avgs,
                            std::vector<at::Tensor> exp_avg_sqs) {
  std::vector<ChunkInfo> chunks;
  for (size_t t = 0; t < params.size(); ++t) {
    TORCH_CHECK(params[t].scalar_type() == at::kFloat);
    auto* p = params[t].data_ptr<float>();
    auto* g = grads[t].data_ptr<float>();
    auto* m = exp_avgs[t].data_ptr<float>();
    auto* v = exp_avg_sqs[t].data_ptr<float>();
    const long n = params[t].numel();
    for (long off = 0; off < n; off += CHUNK) {
      const int len = (int)std::min((long)CHUNK, n - off);
      chunks.push_back({p + off, g + off, m + off, v + off, len});
    }
  }
  auto cpu = at::from_blob(chunks.data(),
                           {(long)(chunks.size() * sizeof(ChunkInfo))},
                           at::TensorOptions().dtype(at::kByte)).clone();
  return cpu.to(params[0].device());
}

void fused_adam_table(at::Tensor table, long n_chunks, at::Tensor hyper,
                      double beta1, double beta2, double eps, double wd) {
  TORCH_CHECK(hyper.is_cuda() && hyper.scalar_type() == at::kFloat &&
              hyper.numel() >= 3);
  const int blocks_x = (CHUNK / 4 + BLOCK - 1) / BLOCK;
  const dim3 grid(std::min(blocks_x, 2048), std::min((int)n_chunks, 64));
  hipLaunchKernelGGL(fused_adam_kernel, grid, dim3(BLOCK), 0,
                     deepof_stream(),
                     reinterpret_cast<const ChunkInfo*>(table.data_ptr()),
                     (int)n_chunks, 0.f, (float)beta1, (float)beta2,
                     (float)eps, (float)wd, 1.f, 1.f,
                     hyper.data_ptr<float>());
}

void fused_adam(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
                std::vector<at::Tensor> exp_avgs,
                std::vector<at::Tensor> exp_avg_sqs, double lr, double beta1,
                double beta2, double eps, double wd, double bias1,
                double bias2) {
  TORCH_CHECK(params.size() == grads.size() &&
              params.size() == exp_avgs.size() &&
              params.size() == exp_avg_sqs.size());
  // thread_local so the host table outlives the async H2D staging
  static thread_local std::vector<ChunkInfo> chunks;
  chunks.clear();
  for (size_t t = 0; t < params.size(); ++t) {
    TORCH_CHECK(params[t].scalar_type() == at::kFloat,
                "fused_adam: fp32 master params only");
    auto* p = params[t].data_ptr<float>();
    auto* g = grads[t].data_ptr<float>();
    auto* m = exp_avgs[t].data_ptr<float>();
    auto* v = exp_avg_sqs[t].data_ptr<float>();
    const long n = params[t].numel();
    for (long off = 0; off < n; off += CHUNK) {
      const int len = (int)std::min((long)CHUNK, n - off);
      chunks.push_back({p + off, g + off, m + off, v + off, len});
    }
  }
  if (chunks.empty()) return;

  auto table = at::empty({(long)(chunks.size() * sizeof(ChunkInfo))},
                         params[0].options().dtype(at::kByte));
  DEEPOF_CHECK_HIP(hipMemcpyAsync(table.data_ptr(), chunks.data(),
                                  chunks.size() * sizeof(ChunkInfo),
                                  hipMemcpyHostToDevice, deepof_stream()));
  // grid.x sized so one chunk saturates; grid.y walks chunks
  const int blocks_x = (CHUNK / 4 + BLOCK - 1) / BLOCK;  // 2048
  const dim3 grid(std::min(blocks_x, 2048),
                  std::min((int)chunks.size(), 64));
  hipLaunchKernelGGL(fused_adam_kernel, grid, dim3(BLOCK), 0,
                     deepof_stream(),
                     reinterpret_cast<const ChunkInfo*>(table.data_ptr()),
                     (int)chunks.size(), (float)lr, (float)beta1,
                     (float)beta2, (float)eps, (float)wd, (float)bias1,
                     (float)bias2, (const float*)nullptr);
}
