// Row-tiled ChebConv for graphs whose activations exceed the LDS budget
// (e.g. 1000-node ER: Ē ≈ 8.5k rows) — BASELINE config 4.
//
// Layout: activations live in the (B, L+1, Ē, 32) acts tensor (global, L2-
// resident at these sizes); each block owns a 64-row tile of one graph and
// stages its X tile + the SpMV result tile in LDS (~42 KB → 3 blocks/CU).
// The dense products run on the same mfma_f32_16x16x4 tiles as the
// LDS-resident kernel.  One launch per layer forward; two per layer
// backward (the dX SpMV needs the full U = δ·W1ᵀ materialized first).

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

#define DEV_INLINE __device__ __forceinline__

constexpr int F = 32;
constexpr int RT = 64;         // rows per tile
constexpr int STRIDE = 33;
typedef float f32x4 __attribute__((ext_vector_type(4)));

// dst_tile(LDS, RT×STRIDE) = rows [r0, r0+RT) of (A · src_global)
DEV_INLINE void spmv_tile(const float* __restrict__ src, float* __restrict__
                          dst, const int* __restrict__ indptr,
                          const int* __restrict__ cols, int r0, int Ee,
                          int tid, int nt) {
    for (int task = tid; task < RT * (F / 4); task += nt) {
        const int rr = task >> 3;
        const int c0 = (task & 7) * 4;
        const int r = r0 + rr;
        float4 acc = {0.f, 0.f, 0.f, 0.f};
        if (r < Ee) {
            for (int a = indptr[r]; a < indptr[r + 1]; ++a) {
                const float4 v = *reinterpret_cast<const float4*>(
                    src + (size_t)cols[a] * F + c0);
                acc.x += v.x; acc.y += v.y; acc.z += v.z; acc.w += v.w;
            }
        }
        float* d = dst + rr * STRIDE + c0;
        d[0] = acc.x; d[1] = acc.y; d[2] = acc.z; d[3] = acc.w;
    }
}

// load a 64-row tile from global (B?,Ee,32) into LDS (RT×STRIDE)
DEV_INLINE void load_tile(const float* __restrict__ src, float* __restrict__
                          dst, int r0, int Ee, int tid, int nt) {
    for (int task = tid; task < RT * (F / 4); task += nt) {
        const int rr = task >> 3;
        const int c0 = (task & 7) * 4;
        float4 v = {0.f, 0.f, 0.f, 0.f};
        if (r0 + rr < Ee)
            v = *reinterpret_cast<const float4*>(
                src + (size_t)(r0 + rr) * F + c0);
        float* d = dst + rr * STRIDE + c0;
        d[0] = v.x; d[1] = v.y; d[2] = v.z; d[3] = v.w;
    }
}

// acc_tile(LDS RT×STRIDE) += src_tile · W (or Wᵀ) — mfma 16×16×4
DEV_INLINE void gemm_tile(const float* __restrict__ src, float* __restrict__
                          dst, const float* __restrict__ W, bool transposeW,
                          int tid) {
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int nw = blockDim.x >> 6;
    const int r_in = lane & 15;
    const int k_in = lane >> 4;
    for (int t = wid; t < (RT / 16) * 2; t += nw) {
        const int mt = t >> 1;
        const int c0 = (t & 1) * 16;
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
        for (int kk = 0; kk < F / 4; ++kk) {
            const int k = kk * 4 + k_in;
            const float a = src[(mt * 16 + r_in) * STRIDE + k];
            const float b = transposeW ? W[(c0 + r_in) * F + k]
                                       : W[k * F + c0 + r_in];
            acc = __builtin_amdgcn_mfma_f32_16x16x4f32(a, b, acc, 0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 4; ++r)
            dst[(mt * 16 + (lane >> 4) * 4 + r) * STRIDE + c0 +
                (lane & 15)] += acc[r];
    }
}

// dW[i][j] += sum over the tile's rows of src[r][i]*delta[r][j] (atomics)
DEV_INLINE void wgrad_tile(const float* __restrict__ src,
                           const float* __restrict__ delta,
                           float* __restrict__ dw_out, int tid) {
    const int lane = tid & 63;
    const int wid = tid >> 6;
    const int nw = blockDim.x >> 6;
    const int k_in = lane >> 4;
    const int c_in = lane & 15;
    for (int t = wid; t < 4; t += nw) {
        const int i0 = (t >> 1) * 16, j0 = (t & 1) * 16;
        f32x4 acc = {0.f, 0.f, 0.f, 0.f};
        for (int kk = 0; kk < RT / 4; ++kk) {
            const int r = kk * 4 + k_in;
            acc = __builtin_amdgcn_mfma_f32_16x16x4f32(
                src[r * STRIDE + i0 + c_in], delta[r * STRIDE + j0 + c_in],
                acc, 0, 0, 0);
        }
#pragma unroll
        for (int r = 0; r < 4; ++r)
            atomicAdd(&dw_out[(i0 + (lane >> 4) * 4 + r) * F + j0
                              + (lane & 15)], acc[r]);
    }
}

// ---- forward, one layer: acts[l+1] = act(X·W0 + (A·X)·W1 + b) -----------
__global__ void cheb_large_fwd_layer(
    const float* __restrict__ x_l,       // (B,Ee,32) = acts[l]
    float* __restrict__ x_next,          // (B,Ee,32) = acts[l+1]
    const float* __restrict__ W,         // (K,32,32) this layer (padded)
    const float* __restrict__ bias,      // (32)
    const int* __restrict__ ext_indptr,  // (B,Ee+1)
    const long* __restrict__ ext_base,
    const int* __restrict__ ext_cols,
    long x_stride,                       // per-graph elements in x_l/x_next
    int Ee, int K, int last) {
    __shared__ __attribute__((aligned(16))) float Xt[RT * STRIDE];
    __shared__ __attribute__((aligned(16))) float Tt[RT * STRIDE];
    __shared__ __attribute__((aligned(16))) float Yt[RT * STRIDE];
    __shared__ float Wl[2 * F * F];
    __shared__ float bl[F];

    const int b = blockIdx.y;
    const int r0 = blockIdx.x * RT;
    const int tid = threadIdx.x, nt = blockDim.x;
    const float* xb = x_l + (size_t)b * x_stride;
    const int* ipt = ext_indptr + (size_t)b * (Ee + 1);
    const int* cls = ext_cols + ext_base[b];

    for (int i = tid; i < K * F * F; i += nt) Wl[i] = W[i];
    for (int i = tid; i < F; i += nt) bl[i] = bias[i];
    for (int i = tid; i < RT * STRIDE; i += nt) Yt[i] = 0.f;
    load_tile(xb, Xt, r0, Ee, tid, nt);
    if (K > 1) spmv_tile(xb, Tt, ipt, cls, r0, Ee, tid, nt);
    __syncthreads();
    gemm_tile(Xt, Yt, Wl, false, tid);
    if (K > 1) gemm_tile(Tt, Yt, Wl + F * F, false, tid);
    __syncthreads();
    float* out = x_next + (size_t)b * x_stride;
    for (int task = tid; task < RT * (F / 4); task += nt) {
        const int rr = task >> 3;
        const int c0 = (task & 7) * 4;
        if (r0 + rr >= Ee) continue;
        float4 v;
#pragma unroll
        for (int q = 0; q < 4; ++q) {
            float y = Yt[rr * STRIDE + c0 + q] + bl[c0 + q];
            y = last ? (y > 0.f ? y : 0.f) : (y > 0.f ? y : 0.2f * y);
            (&v.x)[q] = y;
        }
        *reinterpret_cast<float4*>(out + (size_t)(r0 + rr) * F + c0) = v;
    }
}

// ---- backward stage A (per layer): δpre = δ⊙act'(acts[l+1]);
// db/dW partials; U = δpre·W1ᵀ; delta ← δpre·W0ᵀ --------------------------
__global__ void cheb_large_bwd_a(
    float* __restrict__ delta,           // (B,Ee,32) in: δ, out: δpre·W0ᵀ
    float* __restrict__ U,               // (B,Ee,32) out: δpre·W1ᵀ
    const float* __restrict__ x_l,       // acts[l]
    const float* __restrict__ x_next,    // acts[l+1] (post-act mask)
    const float* __restrict__ W,         // (K,32,32)
    float* __restrict__ dW,              // (B,K,32,32) partials (prezeroed)
    float* __restrict__ db,              // (B,32) partials (prezeroed)
    const int* __restrict__ ext_indptr,
    const long* __restrict__ ext_base,
    const int* __restrict__ ext_cols,
    long x_stride, long dw_stride, long db_stride,
    int Ee, int K, int last, int want_dx) {
    __shared__ __attribute__((aligned(16))) float Dt[RT * STRIDE];
    __shared__ __attribute__((aligned(16))) float Xt[RT * STRIDE];
    __shared__ __attribute__((aligned(16))) float Ot[RT * STRIDE];
    __shared__ float Wl[2 * F * F];

    const int b = blockIdx.y;
    const int r0 = blockIdx.x * RT;
    const int tid = threadIdx.x, nt = blockDim.x;
    const int* ipt = ext_indptr + (size_t)b * (Ee + 1);
    const int* cls = ext_cols + ext_base[b];
    float* del = delta + (size_t)b * Ee * F;
    const float* xa = x_next + (size_t)b * x_stride;
    const float* xl = x_l + (size_t)b * x_stride;

    for (int i = tid; i < K * F * F; i += nt) Wl[i] = W[i];
    // δpre tile (mask applied on load)
    const float slope = last ? 0.f : 0.2f;
    for (int task = tid; task < RT * (F / 4); task += nt) {
        const int rr = task >> 3;
        const int c0 = (task & 7) * 4;
        float4 d = {0.f, 0.f, 0.f, 0.f};
        if (r0 + rr < Ee) {
            d = *reinterpret_cast<const float4*>(
                del + (size_t)(r0 + rr) * F + c0);
            const float4 a = *reinterpret_cast<const float4*>(
                xa + (size_t)(r0 + rr) * F + c0);
            d.x *= a.x > 0.f ? 1.f : slope;
            d.y *= a.y > 0.f ? 1.f : slope;
            d.z *= a.z > 0.f ? 1.f : slope;
            d.w *= a.w > 0.f ? 1.f : slope;
        }
        float* t = Dt + rr * STRIDE + c0;
        t[0] = d.x; t[1] = d.y; t[2] = d.z; t[3] = d.w;
    }
    load_tile(xl, Xt, r0, Ee, tid, nt);
    __syncthreads();

    // db partials (column sums of the tile)
    {
        const int nchunk = nt / F;
        const int j = tid % F, ch = tid / F;
        float acc = 0.f;
        for (int rr = ch; rr < RT; rr += nchunk)
            acc += Dt[rr * STRIDE + j];
        atomicAdd(&db[(size_t)b * db_stride + j], acc);
    }
    wgrad_tile(Xt, Dt, dW + (size_t)b * dw_stride, tid);
    if (K > 1) {
        __syncthreads();
        spmv_tile(xl, Ot, ipt, cls, r0, Ee, tid, nt);   // T1 tile
        __syncthreads();
        wgrad_tile(Ot, Dt, dW + (size_t)b * dw_stride + F * F, tid);
    }
    if (!want_dx) return;
    __syncthreads();

    // U = δpre·W1ᵀ ; newdelta = δpre·W0ᵀ
    if (K > 1) {
        for (int i = tid; i < RT * STRIDE; i += nt) Ot[i] = 0.f;
        __syncthreads();
        gemm_tile(Dt, Ot, Wl + F * F, true, tid);
        __syncthreads();
        float* ub = U + (size_t)b * Ee * F;
        for (int task = tid; task < RT * (F / 4); task += nt) {
            const int rr = task >> 3;
            const int c0 = (task & 7) * 4;
            if (r0 + rr >= Ee) continue;
            float4 v = {Ot[rr * STRIDE + c0], Ot[rr * STRIDE + c0 + 1],
                        Ot[rr * STRIDE + c0 + 2], Ot[rr * STRIDE + c0 + 3]};
            *reinterpret_cast<float4*>(ub + (size_t)(r0 + rr) * F + c0) = v;
        }
    }
    for (int i = tid; i < RT * STRIDE; i += nt) Xt[i] = 0.f;  // reuse as acc
    __syncthreads();
    gemm_tile(Dt, Xt, Wl, true, tid);
    __syncthreads();
    for (int task = tid; task < RT * (F / 4); task += nt) {
        const int rr = task >> 3;
        const int c0 = (task & 7) * 4;
        if (r0 + rr >= Ee) continue;
        float4 v = {Xt[rr * STRIDE + c0], Xt[rr * STRIDE + c0 + 1],
                    Xt[rr * STRIDE + c0 + 2], Xt[rr * STRIDE + c0 + 3]};
        *reinterpret_cast<float4*>(del + (size_t)(r0 + rr) * F + c0) = v;
    }
}

// ---- backward stage B (per layer): delta += A·U --------------------------
__global__ void cheb_large_bwd_b(
    float* __restrict__ delta, const float* __restrict__ U,
    const int* __restrict__ ext_indptr, const long* __restrict__ ext_base,
    const int* __restrict__ ext_cols, int Ee) {
    const int b = blockIdx.y;
    const int r0 = blockIdx.x * RT;
    const int tid = threadIdx.x, nt = blockDim.x;
    const int* ipt = ext_indptr + (size_t)b * (Ee + 1);
    const int* cls = ext_cols + ext_base[b];
    const float* ub = U + (size_t)b * Ee * F;
    float* del = delta + (size_t)b * Ee * F;
    for (int task = tid; task < RT * (F / 4); task += nt) {
        const int rr = task >> 3;
        const int c0 = (task & 7) * 4;
        const int r = r0 + rr;
        if (r >= Ee) continue;
        float4 acc = {0.f, 0.f, 0.f, 0.f};
        for (int a = ipt[r]; a < ipt[r + 1]; ++a) {
            const float4 v = *reinterpret_cast<const float4*>(
                ub + (size_t)cls[a] * F + c0);
            acc.x += v.x; acc.y += v.y; acc.z += v.z; acc.w += v.w;
        }
        float* d = del + (size_t)r * F + c0;
        d[0] += acc.x; d[1] += acc.y; d[2] += acc.z; d[3] += acc.w;
    }
}

}  // namespace

std::vector<torch::Tensor> cheb_large_fwd_hip(
    torch::Tensor x, torch::Tensor W, torch::Tensor bias,
    torch::Tensor ext_indptr, torch::Tensor ext_base,
    torch::Tensor ext_cols) {
    const int B = x.size(0), Ee = x.size(1);
    const int L = W.size(0), K = W.size(1);
    TORCH_CHECK(K <= 2, "cheb_large supports K<=2");
    auto acts = torch::zeros({B, L + 1, (long)Ee, F}, x.options());
    // acts[0][:, :4] = x (rest stays zero)
    acts.select(1, 0).narrow(2, 0, 4).copy_(x);
    auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
    const int tiles = (Ee + RT - 1) / RT;
    for (int l = 0; l < L; ++l) {
        hipLaunchKernelGGL(cheb_large_fwd_layer, dim3(tiles, B), dim3(256),
                           0, stream.stream(),
                           acts.select(1, l).data_ptr<float>(),
                           acts.select(1, l + 1).data_ptr<float>(),
                           W.select(0, l).data_ptr<float>(),
                           bias.select(0, l).data_ptr<float>(),
                           ext_indptr.data_ptr<int>(),
                           ext_base.data_ptr<long>(),
                           ext_cols.data_ptr<int>(),
                           (long)(L + 1) * Ee * F, Ee, K,
                           l == L - 1 ? 1 : 0);
    }
    auto lam = acts.select(1, L).select(2, 0).contiguous();   // (B,Ee)
    return {lam, acts};
}

std::vector<torch::Tensor> cheb_large_bwd_hip(
    torch::Tensor dlam, torch::Tensor acts, torch::Tensor W,
    torch::Tensor ext_indptr, torch::Tensor ext_base,
    torch::Tensor ext_cols) {
    const int B = dlam.size(0), Ee = dlam.size(1);
    const int L = W.size(0), K = W.size(1);
    auto dW = torch::zeros({B, L, K, F, F}, dlam.options());
    auto db = torch::zeros({B, L, F}, dlam.options());
    auto delta = torch::zeros({B, (long)Ee, F}, dlam.options());
    delta.select(2, 0).copy_(dlam);
    auto U = torch::zeros_like(delta);
    auto stream = at::hip::getCurrentHIPStreamMasqueradingAsCUDA();
    const int tiles = (Ee + RT - 1) / RT;
    for (int l = L - 1; l >= 0; --l) {
        const int want_dx = l > 0 ? 1 : 0;
        hipLaunchKernelGGL(cheb_large_bwd_a, dim3(tiles, B), dim3(256), 0,
                           stream.stream(),
                           delta.data_ptr<float>(), U.data_ptr<float>(),
                           acts.select(1, l).data_ptr<float>(),
                           acts.select(1, l + 1).data_ptr<float>(),
                           W.select(0, l).data_ptr<float>(),
                           dW.select(1, l).data_ptr<float>(),
                           db.select(1, l).data_ptr<float>(),
                           ext_indptr.data_ptr<int>(),
                           ext_base.data_ptr<long>(),
                           ext_cols.data_ptr<int>(),
                           (long)(L + 1) * Ee * F,
                           (long)L * K * F * F, (long)L * F,
                           Ee, K, l == L - 1 ? 1 : 0, want_dx);
        if (want_dx && K > 1) {
            hipLaunchKernelGGL(cheb_large_bwd_b, dim3(tiles, B), dim3(256),
                               0, stream.stream(),
                               delta.data_ptr<float>(), U.data_ptr<float>(),
                               ext_indptr.data_ptr<int>(),
                               ext_base.data_ptr<long>(),
                               ext_cols.data_ptr<int>(), Ee);
        }
    }
    return {dW, db};
}
