// Fused optimizer step for gfx950 — SURVEY.md §2.4 K14.
//
// One launch (a single workgroup; the actor has ~3.4k parameters) performs
// the reference's whole update pipeline (gnn_offloading_agent.py:114-121,
// 156-169 + Keras constraint semantics):
//   grad scale (1/batch) → per-tensor clipnorm(1.0) → Adam (eps 1e-7,
//   bias-corrected) → max_norm(1.0) constraints (kernel: per-(fi,fo) norm
//   over the Chebyshev axis; bias: whole-vector norm).
//
// Parameters live in ONE flat buffer (the model's tensors are views), which
// is also the RCCL all-reduce payload — the same flat-buffer layout the DP
// design uses.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

namespace {

// segment descriptor: [offset, size, K, fi, fo]; bias rows have K == 0
__global__ void fused_adam_kernel(
    float* __restrict__ p, float* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v,
    const int* __restrict__ seg, int n_seg, int n_total,
    float scale, float lr, float beta1, float beta2, float eps,
    int* __restrict__ step_dev, int do_constraints) {
    __shared__ float red[256];
    __shared__ float factor;
    __shared__ float bc_s[2];
    const int tid = threadIdx.x;
    const int nt = blockDim.x;
    if (tid == 0) {
        // the step counter lives on device so hipGraph replays advance it
        const float t = (float)(atomicAdd(step_dev, 1) + 1);
        bc_s[0] = 1.0f / (1.0f - powf(beta1, t));
        bc_s[1] = 1.0f / (1.0f - powf(beta2, t));
    }
    __syncthreads();
    const float bc1 = bc_s[0], bc2 = bc_s[1];

    for (int i = tid; i < n_total; i += nt) g[i] *= scale;
    __syncthreads();

    // per-segment clipnorm(1.0), then Adam on the segment
    for (int s = 0; s < n_seg; ++s) {
        const int off = seg[s * 5], size = seg[s * 5 + 1];
        float acc = 0.f;
        for (int i = tid; i < size; i += nt) {
            const float gv = g[off + i];
            acc += gv * gv;
        }
        red[tid] = acc;
        __syncthreads();
        for (int w = nt >> 1; w > 0; w >>= 1) {
            if (tid < w) red[tid] += red[tid + w];
            __syncthreads();
        }
        if (tid == 0) {
            const float nrm = sqrtf(red[0]);
            // non-finite gradient norm (inf/NaN from the 1/(mu-lam) pole
            // in fp32): SKIP this tensor's update entirely — one poisoned
            // step would otherwise write NaN into p/m/v and freeze the
            // policy permanently (observed at seed-dependent rates)
            factor = isfinite(nrm) ? (nrm > 1.0f ? 1.0f / nrm : 1.0f)
                                   : -1.0f;
        }
        __syncthreads();
        const float f = factor;
        if (f >= 0.f) {
            for (int i = tid; i < size; i += nt) {
                const int j = off + i;
                float gv = g[j] * f;
                gv = isfinite(gv) ? gv : 0.f;   // isolated inf/NaN lanes
                const float mn = beta1 * m[j] + (1.f - beta1) * gv;
                const float vn = beta2 * v[j] + (1.f - beta2) * gv * gv;
                m[j] = mn;
                v[j] = vn;
                p[j] -= lr * (mn * bc1) / (sqrtf(vn * bc2) + eps);
            }
        }
        __syncthreads();
    }

    if (!do_constraints) return;
    // Keras max_norm(1.0): kernels — clip the norm over the K axis at each
    // (fi, fo); biases — clip the whole-vector norm
    for (int s = 0; s < n_seg; ++s) {
        const int off = seg[s * 5], size = seg[s * 5 + 1];
        const int K = seg[s * 5 + 2];
        if (K > 0) {
            const int fi = seg[s * 5 + 3], fo = seg[s * 5 + 4];
            const int pairs = fi * fo;
            for (int i = tid; i < pairs; i += nt) {
                float sq = 0.f;
                for (int k = 0; k < K; ++k) {
                    const float w = p[off + k * pairs + i];
                    sq += w * w;
                }
                const float nrm = sqrtf(sq);
                if (nrm > 1.0f) {
                    const float c = 1.0f / nrm;
                    for (int k = 0; k < K; ++k) p[off + k * pairs + i] *= c;
                }
            }
        } else {
            float acc = 0.f;
            for (int i = tid; i < size; i += nt) {
                const float w = p[off + i];
                acc += w * w;
            }
            red[tid] = acc;
            __syncthreads();
            for (int w = nt >> 1; w > 0; w >>= 1) {
                if (tid < w) red[tid] += red[tid + w];
                __syncthreads();
            }
            if (tid == 0) {
                const float nrm = sqrtf(red[0]);
                factor = nrm > 1.0f ? 1.0f / nrm : 1.0f;
            }
            __syncthreads();
            const float f = factor;
            for (int i = tid; i < size; i += nt) p[off + i] *= f;
            __syncthreads();
        }
        __syncthreads();
    }
}

}  // namespace

void fused_adam_hip(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                    torch::Tensor v, torch::Tensor seg, torch::Tensor step,
                    double scale, double lr, double beta1, double beta2,
                    double eps, bool constraints) {
    const int n = p.numel();
    const int n_seg = seg.size(0);
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(fused_adam_kernel, dim3(1), dim3(256), 0,
                       stream.stream(),
                       p.data_ptr<float>(), g.data_ptr<float>(),
                       m.data_ptr<float>(), v.data_ptr<float>(),
                       seg.data_ptr<int>(), n_seg, n,
                       (float)scale, (float)lr, (float)beta1, (float)beta2,
                       (float)eps, step.data_ptr<int>(),
                       constraints ? 1 : 0);
}
