// Batched all-pairs shortest paths (min-plus Floyd–Warshall) for gfx950.
//
// Design (MI355X-first): one workgroup per graph, the whole N×N distance
// matrix resident in LDS (160 KiB/CU ⇒ N ≤ 200 at fp32).  The k-loop runs
// entirely on-chip with one barrier per k; a batch of B graphs fills the
// 256 CUs with B workgroups.  In-place updates are safe: row k and column k
// cannot improve through k itself (d[k][k] = 0), so no thread writes the
// cells another thread reads within an iteration.
//
// Replaces the reference's per-graph networkx Dijkstra
// (util.py:101-110, called twice per method per instance) — the dominant
// CPU hot spot of the reference pipeline.
//
// For N > 200 the host falls back to the tiled global-memory path
// (fw_tiled below): standard 3-phase blocked FW, batched over graphs.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#define LDS_BYTES (160 * 1024)

namespace {

// n_arr: optional per-graph EFFECTIVE node count (inert-padded batches,
// CaseGraph.pad_to): the kernel relaxes only the leading n×n submatrix
// (staged at compact stride n in LDS) — pad nodes are isolated, so their
// +inf rows in global memory are already final.  O(n³) instead of O(N³)
// per graph: a 20-node case padded to 120 does 216× less work.
template <typename T>
__global__ void fw_lds_kernel(T* __restrict__ d, int N,
                              const int* __restrict__ n_arr) {
    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    T* s = reinterpret_cast<T*>(smem_raw);
    T* D = d + (size_t)blockIdx.x * N * N;
    const int n = n_arr ? n_arr[blockIdx.x] : N;
    const int tid = threadIdx.x;
    const int nt = blockDim.x;
    // LDS row stride padded to a float4 multiple with +inf columns: the
    // vectorized relax path then covers EVERY n (min-plus is inert on inf
    // pads), instead of degrading to the scalar path when n % 4 != 0
    const int ns = (sizeof(T) == 4) ? ((n + 3) & ~3) : n;
    for (int c = tid; c < n * ns; c += nt) {
        const int r = c / ns, j = c % ns;
        s[c] = j < n ? D[(size_t)r * N + j] : (T)INFINITY;
    }
    __syncthreads();
    // thread layout: each of 32 lanes per half-wave owns a 4-column group,
    // rows strided across the remaining threads — d[k][j..j+3] loads once
    // per k, d[i][k] broadcasts.
    const int i0 = tid >> 5;
    const int istep = nt >> 5;
    if (sizeof(T) == 4) {
        for (int k = 0; k < n; ++k) {
            for (int j = (tid & 31) * 4; j < n; j += 128) {
                float4 dkj = *reinterpret_cast<const float4*>(
                    reinterpret_cast<const float*>(s) + k * ns + j);
                for (int i = i0; i < n; i += istep) {
                    const float dik =
                        reinterpret_cast<const float*>(s)[i * ns + k];
                    float* row = reinterpret_cast<float*>(s) + i * ns + j;
                    float4 cur = *reinterpret_cast<float4*>(row);
                    cur.x = fminf(cur.x, dik + dkj.x);
                    cur.y = fminf(cur.y, dik + dkj.y);
                    cur.z = fminf(cur.z, dik + dkj.z);
                    cur.w = fminf(cur.w, dik + dkj.w);
                    *reinterpret_cast<float4*>(row) = cur;
                }
            }
            __syncthreads();
        }
    } else {
        for (int k = 0; k < n; ++k) {
            for (int j = tid & 31; j < n; j += 32) {
                const T dkj = s[k * ns + j];
                for (int i = i0; i < n; i += istep) {
                    const T alt = s[i * ns + k] + dkj;
                    if (alt < s[i * ns + j]) s[i * ns + j] = alt;
                }
            }
            __syncthreads();
        }
    }
    for (int c = tid; c < n * n; c += nt)
        D[(size_t)(c / n) * N + (c % n)] = s[(c / n) * ns + (c % n)];
}

// ---- tiled (global-memory) path for large N -------------------------------
// Classic 3-phase blocked FW with TILE=32. Grid z = graph.
constexpr int TILE = 32;

template <typename T>
__device__ inline void fw_tile_body(T* __restrict__ c, const T* __restrict__ a,
                                    const T* __restrict__ b) {
    // c = min(c, a (+) b) over the k dimension of the tile, in LDS
    const int tx = threadIdx.x, ty = threadIdx.y;
    T v = c[ty * TILE + tx];
#pragma unroll
    for (int k = 0; k < TILE; ++k) {
        const T alt = a[ty * TILE + k] + b[k * TILE + tx];
        v = alt < v ? alt : v;
    }
    c[ty * TILE + tx] = v;
}

template <typename T>
__global__ void fw_phase1(T* __restrict__ d, int N, int nb, int kb) {
    __shared__ T tile[TILE * TILE];
    T* D = d + (size_t)blockIdx.z * N * N;
    const int tx = threadIdx.x, ty = threadIdx.y;
    const int r = kb * TILE + ty, c = kb * TILE + tx;
    tile[ty * TILE + tx] = (r < N && c < N) ? D[r * N + c]
                                            : (T)INFINITY;
    __syncthreads();
    for (int k = 0; k < TILE; ++k) {
        const T alt = tile[ty * TILE + k] + tile[k * TILE + tx];
        if (alt < tile[ty * TILE + tx]) tile[ty * TILE + tx] = alt;
        __syncthreads();
    }
    if (r < N && c < N) D[r * N + c] = tile[ty * TILE + tx];
}

template <typename T>
__global__ void fw_phase2(T* __restrict__ d, int N, int nb, int kb) {
    // blockIdx.x: which tile along the strip; blockIdx.y: 0=row strip, 1=col
    __shared__ T piv[TILE * TILE];
    __shared__ T cur[TILE * TILE];
    T* D = d + (size_t)blockIdx.z * N * N;
    int jb = blockIdx.x;
    if (jb >= kb) jb += 1;                  // skip the pivot tile
    if (jb >= nb) return;
    const int tx = threadIdx.x, ty = threadIdx.y;
    const int pr = kb * TILE + ty, pc = kb * TILE + tx;
    piv[ty * TILE + tx] = (pr < N && pc < N) ? D[pr * N + pc] : (T)INFINITY;
    int r, c;
    if (blockIdx.y == 0) { r = kb * TILE + ty; c = jb * TILE + tx; }
    else                 { r = jb * TILE + ty; c = kb * TILE + tx; }
    cur[ty * TILE + tx] = (r < N && c < N) ? D[r * N + c] : (T)INFINITY;
    __syncthreads();
    if (blockIdx.y == 0) {
        for (int k = 0; k < TILE; ++k) {
            const T alt = piv[ty * TILE + k] + cur[k * TILE + tx];
            if (alt < cur[ty * TILE + tx]) cur[ty * TILE + tx] = alt;
            __syncthreads();
        }
    } else {
        for (int k = 0; k < TILE; ++k) {
            const T alt = cur[ty * TILE + k] + piv[k * TILE + tx];
            if (alt < cur[ty * TILE + tx]) cur[ty * TILE + tx] = alt;
            __syncthreads();
        }
    }
    if (r < N && c < N) D[r * N + c] = cur[ty * TILE + tx];
}

template <typename T>
__global__ void fw_phase3(T* __restrict__ d, int N, int nb, int kb) {
    __shared__ T rowt[TILE * TILE];   // tile (ib, kb)
    __shared__ T colt[TILE * TILE];   // tile (kb, jb)
    T* D = d + (size_t)blockIdx.z * N * N;
    int ib = blockIdx.y, jb = blockIdx.x;
    if (ib >= kb) ib += 1;
    if (jb >= kb) jb += 1;
    if (ib >= nb || jb >= nb) return;
    const int tx = threadIdx.x, ty = threadIdx.y;
    int r = ib * TILE + ty, c = kb * TILE + tx;
    rowt[ty * TILE + tx] = (r < N && c < N) ? D[r * N + c] : (T)INFINITY;
    r = kb * TILE + ty; c = jb * TILE + tx;
    colt[ty * TILE + tx] = (r < N && c < N) ? D[r * N + c] : (T)INFINITY;
    __syncthreads();
    r = ib * TILE + ty; c = jb * TILE + tx;
    if (r < N && c < N) {
        T v = D[r * N + c];
#pragma unroll
        for (int k = 0; k < TILE; ++k) {
            const T alt = rowt[ty * TILE + k] + colt[k * TILE + tx];
            v = alt < v ? alt : v;
        }
        D[r * N + c] = v;
    }
}

// ---- TILE=64 fp32 specialization for large N (ER-1000 class) -------------
// Halves the memory-side traffic of the TILE=32 path (4·N³/TILE bytes per
// full FW) and gives each 256-thread block 8× the work per tile triple.
constexpr int T64 = 64;

__global__ void fw64_phase1(float* __restrict__ d, int N, int nb, int kb) {
    __shared__ __attribute__((aligned(16))) float tile[T64 * T64];
    float* D = d + (size_t)blockIdx.z * N * N;
    const int tid = threadIdx.x;                     // 256 threads
    for (int c = tid; c < T64 * T64; c += 256) {
        const int r = kb * T64 + c / T64, cc = kb * T64 + c % T64;
        tile[c] = (r < N && cc < N) ? D[(size_t)r * N + cc] : INFINITY;
    }
    __syncthreads();
    // each thread owns 16 cells (row group of 4 × col group of 4)
    const int ty = tid >> 4, tx = tid & 15;          // 16×16 thread grid
    for (int k = 0; k < T64; ++k) {
        float kj[4], ik[4];
#pragma unroll
        for (int q = 0; q < 4; ++q) {
            kj[q] = tile[k * T64 + tx * 4 + q];
            ik[q] = tile[(ty * 4 + q) * T64 + k];
        }
        __syncthreads();
#pragma unroll
        for (int q = 0; q < 4; ++q)
#pragma unroll
            for (int p = 0; p < 4; ++p) {
                float* cell = tile + (ty * 4 + q) * T64 + tx * 4 + p;
                const float alt = ik[q] + kj[p];
                if (alt < *cell) *cell = alt;
            }
        __syncthreads();
    }
    for (int c = tid; c < T64 * T64; c += 256) {
        const int r = kb * T64 + c / T64, cc = kb * T64 + c % T64;
        if (r < N && cc < N) D[(size_t)r * N + cc] = tile[c];
    }
}

__global__ void fw64_phase2(float* __restrict__ d, int N, int nb, int kb) {
    __shared__ __attribute__((aligned(16))) float piv[T64 * T64];
    __shared__ __attribute__((aligned(16))) float cur[T64 * T64];
    float* D = d + (size_t)blockIdx.z * N * N;
    int jb = blockIdx.x;
    if (jb >= kb) jb += 1;
    if (jb >= nb) return;
    const int tid = threadIdx.x;
    const bool row_strip = blockIdx.y == 0;
    for (int c = tid; c < T64 * T64; c += 256) {
        const int pr = kb * T64 + c / T64, pc = kb * T64 + c % T64;
        piv[c] = (pr < N && pc < N) ? D[(size_t)pr * N + pc] : INFINITY;
        int r, cc;
        if (row_strip) { r = kb * T64 + c / T64; cc = jb * T64 + c % T64; }
        else           { r = jb * T64 + c / T64; cc = kb * T64 + c % T64; }
        cur[c] = (r < N && cc < N) ? D[(size_t)r * N + cc] : INFINITY;
    }
    __syncthreads();
    const int ty = tid >> 4, tx = tid & 15;
    if (row_strip) {
        for (int k = 0; k < T64; ++k) {
            float ik[4], kj[4];
#pragma unroll
            for (int q = 0; q < 4; ++q) {
                ik[q] = piv[(ty * 4 + q) * T64 + k];
                kj[q] = cur[k * T64 + tx * 4 + q];
            }
            __syncthreads();
#pragma unroll
            for (int q = 0; q < 4; ++q)
#pragma unroll
                for (int p = 0; p < 4; ++p) {
                    float* cell = cur + (ty * 4 + q) * T64 + tx * 4 + p;
                    const float alt = ik[q] + kj[p];
                    if (alt < *cell) *cell = alt;
                }
            __syncthreads();
        }
    } else {
        for (int k = 0; k < T64; ++k) {
            float ik[4], kj[4];
#pragma unroll
            for (int q = 0; q < 4; ++q) {
                ik[q] = cur[(ty * 4 + q) * T64 + k];
                kj[q] = piv[k * T64 + tx * 4 + q];
            }
            __syncthreads();
#pragma unroll
            for (int q = 0; q < 4; ++q)
#pragma unroll
                for (int p = 0; p < 4; ++p) {
                    float* cell = cur + (ty * 4 + q) * T64 + tx * 4 + p;
                    const float alt = ik[q] + kj[p];
                    if (alt < *cell) *cell = alt;
                }
            __syncthreads();
        }
    }
    for (int c = tid; c < T64 * T64; c += 256) {
        int r, cc;
        if (row_strip) { r = kb * T64 + c / T64; cc = jb * T64 + c % T64; }
        else           { r = jb * T64 + c / T64; cc = kb * T64 + c % T64; }
        if (r < N && cc < N) D[(size_t)r * N + cc] = cur[c];
    }
}

// phase 3: 256 threads, 4×4 registers per thread over a 64×64 output tile;
// independent min-add chains, no barriers inside the k-loop.
__global__ void fw64_phase3(float* __restrict__ d, int N, int nb, int kb) {
    __shared__ __attribute__((aligned(16))) float rowt[T64 * T64];
    __shared__ __attribute__((aligned(16))) float colt[T64 * T64];
    float* D = d + (size_t)blockIdx.z * N * N;
    int ib = blockIdx.y, jb = blockIdx.x;
    if (ib >= kb) ib += 1;
    if (jb >= kb) jb += 1;
    if (ib >= nb || jb >= nb) return;
    const int tid = threadIdx.x;
    for (int c = tid; c < T64 * T64; c += 256) {
        int r = ib * T64 + c / T64, cc = kb * T64 + c % T64;
        rowt[c] = (r < N && cc < N) ? D[(size_t)r * N + cc] : INFINITY;
        r = kb * T64 + c / T64;
        cc = jb * T64 + c % T64;
        colt[c] = (r < N && cc < N) ? D[(size_t)r * N + cc] : INFINITY;
    }
    __syncthreads();
    const int ty = tid >> 4, tx = tid & 15;
    const int r0 = ib * T64 + ty * 4;
    const int c0 = jb * T64 + tx * 4;
    float4 v[4];
#pragma unroll
    for (int q = 0; q < 4; ++q) {
        v[q] = {INFINITY, INFINITY, INFINITY, INFINITY};
        if (r0 + q < N)
#pragma unroll
            for (int p = 0; p < 4; ++p)
                if (c0 + p < N)
                    (&v[q].x)[p] = D[(size_t)(r0 + q) * N + c0 + p];
    }
#pragma unroll 2
    for (int k = 0; k < T64; ++k) {
        const float4 ckj =
            *reinterpret_cast<const float4*>(colt + k * T64 + tx * 4);
#pragma unroll
        for (int q = 0; q < 4; ++q) {
            const float dik = rowt[(ty * 4 + q) * T64 + k];
            v[q].x = fminf(v[q].x, dik + ckj.x);
            v[q].y = fminf(v[q].y, dik + ckj.y);
            v[q].z = fminf(v[q].z, dik + ckj.z);
            v[q].w = fminf(v[q].w, dik + ckj.w);
        }
    }
#pragma unroll
    for (int q = 0; q < 4; ++q)
        if (r0 + q < N)
#pragma unroll
            for (int p = 0; p < 4; ++p)
                if (c0 + p < N)
                    D[(size_t)(r0 + q) * N + c0 + p] = (&v[q].x)[p];
}

}  // namespace

torch::Tensor floyd_warshall_hip(torch::Tensor w,
                                 c10::optional<torch::Tensor> n_arr) {
    TORCH_CHECK(w.is_cuda() && w.dim() == 3 && w.size(1) == w.size(2),
                "expected (B,N,N) CUDA tensor");
    auto d = w.contiguous().clone();
    const int B = d.size(0), N = d.size(1);
    auto stream = at::cuda::getCurrentCUDAStream();
    const int* np_ = n_arr.has_value() ? n_arr->data_ptr<int>() : nullptr;

    AT_DISPATCH_FLOATING_TYPES(d.scalar_type(), "fw", [&] {
        const int Ns = sizeof(scalar_t) == 4 ? ((N + 3) & ~3) : N;
        const size_t lds = (size_t)N * Ns * sizeof(scalar_t);
        if (lds <= LDS_BYTES) {
            hipLaunchKernelGGL(fw_lds_kernel<scalar_t>, dim3(B), dim3(512),
                               lds, stream.stream(),
                               d.data_ptr<scalar_t>(), N, np_);
        } else if constexpr (std::is_same_v<scalar_t, float>) {
            // fp32 large-N: TILE=64 (half the fabric traffic of TILE=32,
            // 8x the work per block; ER-1000 class)
            const int nb = (N + T64 - 1) / T64;
            for (int kb = 0; kb < nb; ++kb) {
                hipLaunchKernelGGL(fw64_phase1, dim3(1, 1, B), dim3(256), 0,
                                   stream.stream(), d.data_ptr<float>(),
                                   N, nb, kb);
                if (nb > 1) {
                    hipLaunchKernelGGL(fw64_phase2, dim3(nb - 1, 2, B),
                                       dim3(256), 0, stream.stream(),
                                       d.data_ptr<float>(), N, nb, kb);
                    hipLaunchKernelGGL(fw64_phase3,
                                       dim3(nb - 1, nb - 1, B), dim3(256),
                                       0, stream.stream(),
                                       d.data_ptr<float>(), N, nb, kb);
                }
            }
        } else {
            const int nb = (N + TILE - 1) / TILE;
            dim3 thr(TILE, TILE);
            for (int kb = 0; kb < nb; ++kb) {
                hipLaunchKernelGGL(fw_phase1<scalar_t>, dim3(1, 1, B), thr, 0,
                                   stream.stream(), d.data_ptr<scalar_t>(),
                                   N, nb, kb);
                if (nb > 1) {
                    hipLaunchKernelGGL(fw_phase2<scalar_t>,
                                       dim3(nb - 1, 2, B), thr, 0,
                                       stream.stream(),
                                       d.data_ptr<scalar_t>(), N, nb, kb);
                    hipLaunchKernelGGL(fw_phase3<scalar_t>,
                                       dim3(nb - 1, nb - 1, B), thr, 0,
                                       stream.stream(),
                                       d.data_ptr<scalar_t>(), N, nb,
                                       kb);
                }
            }
        }
    });
    return d;
}
