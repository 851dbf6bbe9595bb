// Fused episode kernels for gfx950: one workgroup per graph, queueing state
// staged in LDS.  These replace the torch gather/scatter/index chains of the
// engine's decision, routing-walk and evaluation stages (profiled at ~35% of
// step GPU time plus the per-hop host sync of the walk loop).
//
// Reference semantics (clean-room): offloading_v3.py:388-550.
// Oracle: multihop_offload_amd/engine.py torch path (tests/test_gpu.py).
//
// ε-greedy exploration and the softmax ("prob") sampling mode run INSIDE
// decide_kernel with a counter-based stateless RNG (splitmix64 over
// (seed, step counter, b, j)): `explore` is read from a device scalar and
// the counter from a device tensor the engine bumps once per step, so the
// whole decision stage is a single launch and hipGraph capture replays
// with fresh randomness.

#include <hip/hip_runtime.h>
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

namespace {

#define DEV_INLINE __device__ __forceinline__

DEV_INLINE float fmax1(float a, float b) { return a > b ? a : b; }

// splitmix64 — stateless counter-based RNG (statistically strong per-call
// avalanche; each (seed, counter, b, j) tuple is an independent draw)
DEV_INLINE unsigned long long mix64(unsigned long long z) {
    z += 0x9E3779B97F4A7C15ull;
    z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
    z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
    return z ^ (z >> 31);
}

DEV_INLINE float u01(unsigned long long h) {
    // top 24 bits → [0,1) float
    return (float)(h >> 40) * (1.0f / 16777216.0f);
}

// ---------------------------------------------------------------------------
// decide: per-job offloading costs + argmin choice (offloading_v3.py:401-422)
// + in-kernel ε-greedy / softmax sampling (:416-422).
// grid.x = B, threads loop jobs; each thread scans the S servers serially.
// ---------------------------------------------------------------------------
__global__ void decide_kernel(
    const float* __restrict__ sp,       // (B,N,N) diag=0
    const float* __restrict__ hop,      // (B,N,N)
    const float* __restrict__ uds,      // (B,N)   inf at relays
    const int* __restrict__ servers,    // (B,S)   pad -1 (tail-contiguous)
    const long* __restrict__ src,       // (B,J)
    const bool* __restrict__ mask,      // (B,J)
    const float* __restrict__ ul,       // (B,J)
    const float* __restrict__ dl,       // (B,J)
    const float* __restrict__ explore,  // device scalar | nullptr
    const long* __restrict__ rng,       // (2,): seed, step counter | nullptr
    long* __restrict__ dst_out,         // (B,J)
    bool* __restrict__ islocal_out,     // (B,J)
    int prob, int N, int J, int S) {
    const int b = blockIdx.x;
    const float* spb = sp + (size_t)b * N * N;
    const float* hpb = hop + (size_t)b * N * N;
    const float* udsb = uds + (size_t)b * N;
    const int* srvb = servers + (size_t)b * S;
    const float eps = explore ? *explore : 0.0f;
    const unsigned long long key =
        rng ? mix64((unsigned long long)rng[0]
                    ^ mix64((unsigned long long)rng[1])) : 0ull;
    int nS = 0;
    for (int s = 0; s < S; ++s) nS += (srvb[s] >= 0);
    for (int j = threadIdx.x; j < J; j += blockDim.x) {
        const size_t bj = (size_t)b * J + j;
        const int s0 = (int)src[bj];
        if (!mask[bj]) { dst_out[bj] = s0; islocal_out[bj] = true; continue; }
        const float ulj = ul[bj], dlj = dl[bj];
        const float local = udsb[s0] * ulj;
        int best_s;
        if (prob) {
            // softmax over raw costs (the reference's prob mode quirk:
            // HIGH-cost servers get HIGH probability — util.py:113-116);
            // two passes: max, then inverse-CDF walk with one uniform
            float cmax = local < 1e30f ? local : 1e30f;
            for (int s = 0; s < S; ++s) {
                const int sv = srvb[s];
                if (sv < 0) continue;
                const float spv = spb[(size_t)s0 * N + sv];
                const float hpv = hpb[(size_t)s0 * N + sv];
                float c = fmax1(spv * ulj, hpv) + fmax1(spv * dlj, hpv)
                          + fmax1(udsb[sv] * ulj, 1.0f);
                if (isfinite(c)) { c = c < 1e30f ? c : 1e30f;
                                   cmax = c > cmax ? c : cmax; }
            }
            float psum = 0.f;
            for (int s = 0; s < S; ++s) {
                const int sv = srvb[s];
                if (sv < 0) continue;
                const float spv = spb[(size_t)s0 * N + sv];
                const float hpv = hpb[(size_t)s0 * N + sv];
                float c = fmax1(spv * ulj, hpv) + fmax1(spv * dlj, hpv)
                          + fmax1(udsb[sv] * ulj, 1.0f);
                psum += isfinite(c) ? __expf((c < 1e30f ? c : 1e30f) - cmax)
                                    : 0.f;
            }
            const float ploc = isfinite(local)
                ? __expf((local < 1e30f ? local : 1e30f) - cmax) : 0.f;
            psum += ploc;
            const float r = u01(mix64(key ^ (unsigned long long)bj)) * psum;
            float acc = 0.f;
            best_s = -1;                     // fallthrough: local
            for (int s = 0; s < S; ++s) {
                const int sv = srvb[s];
                if (sv < 0) continue;
                const float spv = spb[(size_t)s0 * N + sv];
                const float hpv = hpb[(size_t)s0 * N + sv];
                float c = fmax1(spv * ulj, hpv) + fmax1(spv * dlj, hpv)
                          + fmax1(udsb[sv] * ulj, 1.0f);
                acc += isfinite(c) ? __expf((c < 1e30f ? c : 1e30f) - cmax)
                                   : 0.f;
                if (r < acc) { best_s = s; break; }
            }
        } else {
            // torch argmin tie-break = first index; the cost vector is
            // [server_0 .. server_{S-1}, local]: first-min among servers via
            // strict <, and local wins only if strictly below every server.
            float best = INFINITY;
            best_s = -1;
            for (int s = 0; s < S; ++s) {
                const int sv = srvb[s];
                if (sv < 0) continue;
                const float spv = spb[(size_t)s0 * N + sv];
                const float hpv = hpb[(size_t)s0 * N + sv];
                const float c = fmax1(spv * ulj, hpv) + fmax1(spv * dlj, hpv)
                              + fmax1(udsb[sv] * ulj, 1.0f);
                if (c < best) { best = c; best_s = s; }
            }
            if (local < best) best_s = -1;      // -1 = local
        }
        if (eps > 0.f) {
            const unsigned long long h1 =
                mix64(key ^ (0x517CC1B727220A95ull + bj));
            if (u01(h1) < eps) {
                // uniform over the nS valid servers + local
                const int rc = (int)(u01(mix64(h1)) * (float)(nS + 1));
                best_s = rc >= nS ? -1 : rc;
            }
        }
        dst_out[bj] = best_s < 0 ? s0 : srvb[best_s];
        islocal_out[bj] = best_s < 0;
    }
}

// ---------------------------------------------------------------------------
// walk_eval: greedy next-hop walk + load accumulation + contention fixed
// point + per-job empirical delays (offloading_v3.py:441-550).
// grid.x = B; one workgroup per graph; LDS: lam[E], mu[E], nb[E], sload[N].
//
// unit_mtx determinism: several jobs can traverse the same link (or share a
// destination) with DIFFERENT per-job fallback units when congested; the
// oracle's Python loop makes the LAST job's value win.  Stage 3 reproduces
// that deterministically under parallel execution by packing
// (job_index+1) << 32 | float_bits(unit) into a u64 scratch cell with
// atomicMax (units are finite non-negative, so the job index dominates),
// then unpacking after a barrier — block b owns slice b, so no cross-block
// coherence is needed.
// ---------------------------------------------------------------------------
__global__ void walk_eval_kernel(
    const float* __restrict__ sp,        // (B,N,N)
    const long* __restrict__ src,        // (B,J)
    const long* __restrict__ dstv,       // (B,J)
    const bool* __restrict__ mask,       // (B,J)
    const float* __restrict__ rate,      // (B,J)
    const float* __restrict__ ul,        // (B,J)
    const float* __restrict__ dl,        // (B,J)
    const int* __restrict__ adj_indptr,  // (B,N+1)
    const int* __restrict__ adj_idx,     // (B,2E)
    const int* __restrict__ adj_link,    // (B,2E)
    const int* __restrict__ conf_indptr, // (B,E+1) into conf_cols[conf_base]
    const long* __restrict__ conf_base,  // (B+1)
    const int* __restrict__ conf_cols,   // flat, local link ids
    const float* __restrict__ rates,     // (B,E)
    const float* __restrict__ bw,        // (B,N)
    const int* __restrict__ edges,       // (B,E,2)
    int* __restrict__ route_links,       // (B,J,H) out, -1 pad
    int* __restrict__ nhop,              // (B,J) out
    float* __restrict__ delay_emp,       // (B,J) out (nan for padded jobs)
    float* __restrict__ unit_mtx,        // (B,N,N) out (prezeroed)
    bool* __restrict__ written,          // (B,N,N) out (prezeroed)
    unsigned long long* __restrict__ upack,  // (B,N,N) scratch
    int* __restrict__ overflow,          // (B) out
    const float* __restrict__ T_arr,     // (B)
    const int* __restrict__ E_arr,       // (B) real link counts
    const int* __restrict__ n_arr,       // (B) real node counts (padding)
    int N, int E, int J, int H, int fp_iters) {
    extern __shared__ __attribute__((aligned(16))) char smem_raw[];
    float* lam = reinterpret_cast<float*>(smem_raw);          // E
    float* mu = lam + E;                                      // E
    float* busy = mu + E;                                     // E
    float* sload = busy + E;                                  // N

    const int b = blockIdx.x;
    const int tid = threadIdx.x;
    const int nt = blockDim.x;
    const float* spb = sp + (size_t)b * N * N;
    const int* aip = adj_indptr + (size_t)b * (N + 1);
    const int* aix = adj_idx + (size_t)b * 2 * E;
    const int* alk = adj_link + (size_t)b * 2 * E;
    const float* ratesb = rates + (size_t)b * E;
    const int* cip = conf_indptr + (size_t)b * (E + 1);
    const int* ccols = conf_cols + conf_base[b];
    const float T = T_arr[b];
    unsigned long long* upk = upack + (size_t)b * N * N;
    // per-graph effective sizes: padded link slots / inert pad nodes are
    // isolated and never touched by routes, so every E/N-length loop below
    // runs on the real extent only (ragged-E batches, pad_to mixed-N)
    const int Eb = E_arr[b];
    const int nb_ = n_arr[b];

    for (int e = tid; e < Eb; e += nt) lam[e] = 0.0f;
    for (int n = tid; n < nb_; n += nt) sload[n] = 0.0f;
    for (int i = tid; i < nb_ * nb_; i += nt)
        upk[(i / nb_) * N + (i % nb_)] = 0ull;
    __syncthreads();

    // ---- stage 1: walk + load accumulation -------------------------------
    for (int j = tid; j < J; j += nt) {
        const size_t bj = (size_t)b * J + j;
        if (!mask[bj]) {
            nhop[bj] = 0;
            for (int h = 0; h < H; ++h)
                route_links[bj * H + h] = -1;
            continue;
        }
        const int d = (int)dstv[bj];
        int node = (int)src[bj];
        const float add = (ul[bj] + dl[bj]) * rate[bj];
        int h = 0;
        while (node != d && h < H) {
            const int lo = aip[node], hi = aip[node + 1];
            float bestv = INFINITY;
            int bestn = -1, bestl = -1;
            for (int a = lo; a < hi; ++a) {
                const int nb2 = aix[a];
                const float v = spb[(size_t)nb2 * N + d];
                if (v < bestv) { bestv = v; bestn = nb2; bestl = alk[a]; }
            }
            route_links[((size_t)b * J + j) * H + h] = bestl;
            atomicAdd(&lam[bestl], add);
            node = bestn;
            ++h;
        }
        if (node != d) atomicAdd(&overflow[b], 1);
        nhop[bj] = h;
        for (int h2 = h; h2 < H; ++h2)       // outputs arrive torch::empty
            route_links[bj * H + h2] = -1;
        atomicAdd(&sload[d], ul[bj] * rate[bj]);
    }
    __syncthreads();

    // ---- stage 2: contention fixed point (offloading_v3.py:500-506) ------
    for (int e = tid; e < Eb; e += nt) {
        const float deg = (float)(cip[e + 1] - cip[e]);
        mu[e] = ratesb[e] / (deg + 1.0f);
    }
    __syncthreads();
    for (int it = 0; it < fp_iters; ++it) {
        for (int e = tid; e < Eb; e += nt) {
            const float r = lam[e] / mu[e];
            busy[e] = r < 0.f ? 0.f : (r > 1.f ? 1.f : r);
        }
        __syncthreads();
        for (int e = tid; e < Eb; e += nt) {
            float nbv = 0.0f;
            for (int a = cip[e]; a < cip[e + 1]; ++a) nbv += busy[ccols[a]];
            mu[e] = ratesb[e] / (1.0f + nbv);
        }
        __syncthreads();
    }

    // ---- stage 3: per-job empirical delays (offloading_v3.py:522-549) ----
    const float* bwb = bw + (size_t)b * N;
    const int* edg = edges + (size_t)b * E * 2;
    for (int j = tid; j < J; j += nt) {
        const size_t bj = (size_t)b * J + j;
        if (!mask[bj]) { delay_emp[bj] = NAN; continue; }
        const float ulj = ul[bj], dlj = dl[bj];
        const float tot = ulj + dlj;
        const float nh = (float)nhop[bj];
        const unsigned long long jtag = ((unsigned long long)(j + 1)) << 32;
        float acc = 0.0f;
        for (int h = 0; h < nhop[bj]; ++h) {
            const int l = route_links[((size_t)b * J + j) * H + h];
            if (l < 0) break;
            const float gap = mu[l] - lam[l];
            const float unit = gap > 0.f ? 1.0f / gap
                                         : T * lam[l] / (tot * mu[l]);
            const int u = edg[l * 2], v = edg[l * 2 + 1];
            const unsigned long long pk = jtag | (unsigned long long)
                __float_as_uint(unit);
            atomicMax(&upk[(size_t)u * N + v], pk);
            atomicMax(&upk[(size_t)v * N + u], pk);
            acc += fmax1(ulj * unit, nh) + fmax1(dlj * unit, nh);
        }
        const int d = (int)dstv[bj];
        const float sgap = bwb[d] - sload[d];
        const float sunit = sgap > 0.f ? 1.0f / sgap
                                       : T * sload[d] / (ulj * bwb[d]);
        atomicMax(&upk[(size_t)d * N + d],
                  jtag | (unsigned long long)__float_as_uint(sunit));
        acc += fmax1(ulj * sunit, 1.0f);
        delay_emp[bj] = acc;
    }
    __syncthreads();
    // ---- stage 3b: unpack last-job-wins units into unit_mtx/written ------
    float* um = unit_mtx + (size_t)b * N * N;
    bool* wm = written + (size_t)b * N * N;
    // outputs arrive torch::empty: every cell is written exactly once
    // (pad region → 0/false), replacing two fill launches per call
    for (int i = tid; i < N * N; i += nt) {
        const int r = i / N, c = i % N;
        const unsigned long long pk =
            (r < nb_ && c < nb_) ? upk[i] : 0ull;
        um[i] = pk ? __uint_as_float((unsigned int)(pk & 0xffffffffull))
                   : 0.f;
        wm[i] = pk != 0ull;
    }
}

}  // namespace

std::vector<torch::Tensor> decide_hip(
    torch::Tensor sp, torch::Tensor hop, torch::Tensor uds,
    torch::Tensor servers, torch::Tensor src, torch::Tensor mask,
    torch::Tensor ul, torch::Tensor dl,
    c10::optional<torch::Tensor> explore, c10::optional<torch::Tensor> rng,
    long prob) {
    const int B = sp.size(0), N = sp.size(1);
    const int J = src.size(1), S = servers.size(1);
    auto dst = torch::empty({B, J}, src.options());
    auto islocal = torch::empty({B, J}, mask.options());
    const float* ep = explore.has_value()
        ? explore->data_ptr<float>() : nullptr;
    const long* rp = rng.has_value() ? rng->data_ptr<long>() : nullptr;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(decide_kernel, dim3(B), dim3(256), 0, stream.stream(),
                       sp.data_ptr<float>(), hop.data_ptr<float>(),
                       uds.data_ptr<float>(),
                       servers.data_ptr<int>(), src.data_ptr<long>(),
                       mask.data_ptr<bool>(), ul.data_ptr<float>(),
                       dl.data_ptr<float>(), ep, rp,
                       dst.data_ptr<long>(),
                       islocal.data_ptr<bool>(), (int)prob, N, J, S);
    return {dst, islocal};
}

std::vector<torch::Tensor> walk_eval_hip(
    torch::Tensor sp, torch::Tensor src, torch::Tensor dst,
    torch::Tensor mask, torch::Tensor rate, torch::Tensor ul,
    torch::Tensor dl, torch::Tensor adj_indptr, torch::Tensor adj_idx,
    torch::Tensor adj_link, torch::Tensor conf_indptr,
    torch::Tensor conf_base, torch::Tensor conf_cols, torch::Tensor rates,
    torch::Tensor bw, torch::Tensor edges, torch::Tensor T_arr,
    torch::Tensor E_arr, torch::Tensor n_arr, long H,
    long fp_iters) {
    const int B = sp.size(0), N = sp.size(1);
    const int J = src.size(1), E = rates.size(1);
    auto opts_i = adj_indptr.options();
    auto opts_f = sp.options();
    // all outputs fully initialized in-kernel (no fill launches)
    auto route_links = torch::empty({B, J, H}, opts_i.dtype(torch::kInt32));
    auto nhop = torch::empty({B, J}, opts_i.dtype(torch::kInt32));
    auto delay_emp = torch::empty({B, J}, opts_f);
    auto unit_mtx = torch::empty({B, N, N}, opts_f);
    auto written = torch::empty({B, N, N}, opts_f.dtype(torch::kBool));
    // u64 scratch for deterministic last-job-wins unit writes (the kernel
    // zeroes its own slice — empty, not zeros)
    auto upack = torch::empty({B, N, N}, opts_f.dtype(torch::kInt64));
    auto overflow = torch::zeros({B}, opts_i.dtype(torch::kInt32));
    const size_t lds = sizeof(float) * (3 * (size_t)E + N);
    TORCH_CHECK(lds <= 160 * 1024, "graph too large for LDS walk_eval");
    // large graphs ship few workgroups (one per graph): widen them so the
    // E-length loops and the N*N unpack keep more lanes busy per CU
    const int threads = (E >= 1500 || N >= 500) ? 1024 : 256;
    auto stream = at::cuda::getCurrentCUDAStream();
    hipLaunchKernelGGL(walk_eval_kernel, dim3(B), dim3(threads), lds,
                       stream.stream(),
                       sp.data_ptr<float>(), src.data_ptr<long>(),
                       dst.data_ptr<long>(), mask.data_ptr<bool>(),
                       rate.data_ptr<float>(), ul.data_ptr<float>(),
                       dl.data_ptr<float>(),
                       adj_indptr.data_ptr<int>(), adj_idx.data_ptr<int>(),
                       adj_link.data_ptr<int>(),
                       conf_indptr.data_ptr<int>(),
                       conf_base.data_ptr<long>(),
                       conf_cols.data_ptr<int>(), rates.data_ptr<float>(),
                       bw.data_ptr<float>(), edges.data_ptr<int>(),
                       route_links.data_ptr<int>(), nhop.data_ptr<int>(),
                       delay_emp.data_ptr<float>(),
                       unit_mtx.data_ptr<float>(), written.data_ptr<bool>(),
                       reinterpret_cast<unsigned long long*>(
                           upack.data_ptr<long>()),
                       overflow.data_ptr<int>(),
                       T_arr.data_ptr<float>(), E_arr.data_ptr<int>(),
                       n_arr.data_ptr<int>(), N, E, J, (int)H,
                       (int)fp_iters);
    return {route_links, nhop, delay_emp, unit_mtx, written, overflow};
}
