// Fused BatchNorm(+Add)(+ReLU) for NHWC bf16 activations on gfx950.
//
// MIOpen's spatial-BN kernels plus the separate residual-add / ReLU /
// ReLU-backward elementwise kernels account for ~half the non-conv backbone
// time (rocprof: MIOpenBatchNormBwdSpatialDX + DScaleDBias + FwdTrain* +
// CUDAFunctor_add + threshold ~ 25% of the step).  These kernels fuse the
// whole tail of a conv (normalize + residual add + ReLU) into one
// vectorized pass per direction, per the CDNA guide's "fuse elementwise/
// normalisation/activation work into the producing kernel" rule.
//
// Layout: activations are [M, C] row-major views of NHWC tensors
// (M = N*H*W), bf16, C % 8 == 0; statistics and parameters are fp32.
// Each lane owns 8 consecutive channels (one 16-byte vector load), so a
// wave reads 1 KiB contiguous — fully coalesced.  Per-channel reductions
// are deterministic two-stage: register partials -> LDS combine -> one
// partial row per block -> ILP-unrolled stage-2 sum (no atomics).

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <ATen/hip/HIPContext.h>
#include <hip/hip_bf16.h>

using bf16 = __hip_bfloat16;
using short8 = __attribute__((ext_vector_type(8))) short;

__device__ __forceinline__ float bf2f(short s) {
    union { float f; unsigned u; } cvt;
    cvt.u = ((unsigned)(unsigned short)s) << 16;
    return cvt.f;
}

__device__ __forceinline__ short f2bf(float f) {
    union { float f; unsigned u; } cvt;
    cvt.f = f;
    unsigned u = cvt.u;
    // round-to-nearest-even (matches PyTorch's float->bf16 cast)
    unsigned rounding_bias = 0x7FFF + ((u >> 16) & 1);
    return (short)((u + rounding_bias) >> 16);
}

#define CHECK_BN(x) TORCH_CHECK(x.is_cuda() && x.is_contiguous(), #x " must be contiguous on device")

// ---------------------------------------------------------------------------
// forward: per-channel sums (training stats)
// ---------------------------------------------------------------------------

// Two-stage deterministic reduction (no atomics): stage 1 blocks cover
// rows_per_iter = 256/(C/8) rows per iteration (all 256 threads active for
// any C), accumulate per-thread partials in registers, tree-combine
// row-groups through LDS, and write ONE partial row [2C] per block; a tiny
// stage-2 kernel sums the partial rows.  Wave lanes read consecutive rows'
// consecutive 16-byte chunks -> fully coalesced.

// LDS combine helper: each thread holds 16 floats (8 s, 8 q) for channel
// chunk (tid % tpr); writes lds[tid][16] then 256 threads re-reduce the
// rpi row-groups serially per output element. Emits block partial [2C].
__device__ __forceinline__
void bn_block_partial(float (&s)[8], float (&q)[8], float* lds, int tpr,
                      float* __restrict__ partial, int C) {
    // 17-word per-thread stride: 17 is coprime with the 32-bank write
    // modulus, so the 16 scalar stores per thread are conflict-free
    // (stride 16 measured ~100M SQ_LDS_BANK_CONFLICT cycles per bench)
    const int tid = threadIdx.x;
    #pragma unroll
    for (int i = 0; i < 8; ++i) {
        lds[tid * 17 + i] = s[i];
        lds[tid * 17 + 8 + i] = q[i];
    }
    __syncthreads();
    const int rpi = 256 / tpr;
    for (int e = tid; e < tpr * 16; e += 256) {
        const int chunk = e / 16;
        const int comp = e % 16;
        float acc = 0.f;
        for (int r = 0; r < rpi; ++r)
            acc += lds[(r * tpr + chunk) * 17 + comp];
        const int c = chunk * 8 + (comp & 7);
        partial[(comp < 8 ? c : C + c)] = acc;
    }
}

__global__ __launch_bounds__(256)
void bn_stats_kernel(const short* __restrict__ x,
                     float* __restrict__ partials,  // [gridDim.x, 2C]
                     long M, int C) {
    __shared__ __attribute__((aligned(16))) float lds[256 * 17];
    const int tpr = C / 8;
    float s[8] = {0}, q[8] = {0};
    if (tpr >= 256) {                       // channel-split path
        const int c8 = (blockIdx.y * 256 + threadIdx.x) * 8;
        float* partial = partials + ((long)blockIdx.y * gridDim.x
                                     + blockIdx.x) * 2 * C;
        if (c8 < C) {
            // 4 independent 16B loads in flight per iteration: one load per
            // loop trip leaves the memory pipe under-occupied (stats
            // measured 2.1 TB/s vs the 2-op apply kernel's 4.9 TB/s)
            const long st = gridDim.x;
            long m = blockIdx.x;
            for (; m + 3 * st < M; m += 4 * st) {
                short8 v[4];
                #pragma unroll
                for (int u = 0; u < 4; ++u)
                    v[u] = *reinterpret_cast<const short8*>(
                        x + (m + u * st) * C + c8);
                #pragma unroll
                for (int u = 0; u < 4; ++u)
                    #pragma unroll
                    for (int i = 0; i < 8; ++i) {
                        const float f = bf2f(v[u][i]);
                        s[i] += f; q[i] += f * f;
                    }
            }
            for (; m < M; m += st) {
                const short8 v = *reinterpret_cast<const short8*>(x + m * C + c8);
                #pragma unroll
                for (int i = 0; i < 8; ++i) {
                    const float f = bf2f(v[i]);
                    s[i] += f; q[i] += f * f;
                }
            }
            #pragma unroll
            for (int i = 0; i < 8; ++i) {
                partial[c8 + i] = s[i];
                partial[C + c8 + i] = q[i];
            }
        }
        return;
    }
    const int rpi = 256 / tpr;
    const int rsub = threadIdx.x / tpr;
    const int c8 = (threadIdx.x % tpr) * 8;
    if (rsub < rpi) {
        const long st = (long)gridDim.x * rpi;
        long m = (long)blockIdx.x * rpi + rsub;
        for (; m + 3 * st < M; m += 4 * st) {
            short8 v[4];
            #pragma unroll
            for (int u = 0; u < 4; ++u)
                v[u] = *reinterpret_cast<const short8*>(
                    x + (m + u * st) * C + c8);
            #pragma unroll
            for (int u = 0; u < 4; ++u)
                #pragma unroll
                for (int i = 0; i < 8; ++i) {
                    const float f = bf2f(v[u][i]);
                    s[i] += f; q[i] += f * f;
                }
        }
        for (; m < M; m += st) {
            const short8 v = *reinterpret_cast<const short8*>(x + m * C + c8);
            #pragma unroll
            for (int i = 0; i < 8; ++i) {
                const float f = bf2f(v[i]);
                s[i] += f; q[i] += f * f;
            }
        }
    }
    bn_block_partial(s, q, lds, tpr, partials + (long)blockIdx.x * 2 * C, C);
}

// stage 2: sums[e] = sum over nb partial rows
__global__ __launch_bounds__(256)
void bn_partial_sum_kernel(const float* __restrict__ partials,
                           float* __restrict__ sums, int nb, int C2) {
    const int e = blockIdx.x * 256 + threadIdx.x;
    if (e >= C2) return;
    // 8 independent accumulators keep 8 loads in flight (the serial form
    // was 1280 dependent L2 round trips = ~270 us per call)
    float a[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    int b = 0;
    for (; b + 8 <= nb; b += 8) {
        #pragma unroll
        for (int i = 0; i < 8; ++i)
            a[i] += partials[(long)(b + i) * C2 + e];
    }
    for (; b < nb; ++b)
        a[0] += partials[(long)b * C2 + e];
    sums[e] = ((a[0] + a[1]) + (a[2] + a[3])) + ((a[4] + a[5]) + (a[6] + a[7]));
}

// stage 2, block-per-element form: with nb up to 1024 rows the thread-per-
// element kernel leaves mid-size layers (C<=512) with <=256 active threads
// doing 1024 serial-ish column loads — 30-40 us while the data is 2 MB.
// One block per element + fixed-tree LDS reduction is deterministic and
// fully parallel (adjacent blocks read adjacent columns -> L2-coalesced).
__global__ __launch_bounds__(256)
void bn_partial_sum_block_kernel(const float* __restrict__ partials,
                                 float* __restrict__ sums, int nb, int C2) {
    const int e = blockIdx.x;
    const int tid = threadIdx.x;
    float a = 0.f;
    for (int b = tid; b < nb; b += 256)
        a += partials[(long)b * C2 + e];
    __shared__ float lds[256];
    lds[tid] = a;
    __syncthreads();
    #pragma unroll
    for (int s = 128; s > 0; s >>= 1) {
        if (tid < s) lds[tid] += lds[tid + s];
        __syncthreads();
    }
    if (tid == 0) sums[e] = lds[0];
}

// dispatch: many partial rows + few channels -> block-per-element
static inline void launch_partial_sum(const float* partials, float* sums,
                                      int nb, int C2, hipStream_t stream) {
    if (nb >= 64 && C2 <= 4096) {
        hipLaunchKernelGGL(bn_partial_sum_block_kernel, dim3(C2), dim3(256),
                           0, stream, partials, sums, nb, C2);
    } else {
        hipLaunchKernelGGL(bn_partial_sum_kernel, dim3((C2 + 255) / 256),
                           dim3(256), 0, stream, partials, sums, nb, C2);
    }
}

// finalize: mean/rstd, running-stat update, scale/shift for the apply pass
__global__ __launch_bounds__(256)
void bn_finalize_kernel(const float* __restrict__ sums,
                        const float* __restrict__ weight,
                        const float* __restrict__ bias,
                        float* __restrict__ running_mean,
                        float* __restrict__ running_var,
                        float* __restrict__ save_mean,
                        float* __restrict__ save_rstd,
                        float* __restrict__ scale_shift,  // [2C]
                        long M, int C, float momentum, float eps,
                        int update_running) {
    const int c = blockIdx.x * 256 + threadIdx.x;
    if (c >= C) return;
    const float mean = sums[c] / (float)M;
    float var = sums[C + c] / (float)M - mean * mean;
    var = fmaxf(var, 0.f);
    const float rstd = rsqrtf(var + eps);
    save_mean[c] = mean;
    save_rstd[c] = rstd;
    if (update_running) {
        const float unbiased = (M > 1) ? var * (float)M / (float)(M - 1) : var;
        running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
        running_var[c] = (1.f - momentum) * running_var[c] + momentum * unbiased;
    }
    const float sc = weight[c] * rstd;
    scale_shift[c] = sc;
    scale_shift[C + c] = bias[c] - mean * sc;
}

// eval-mode scale/shift straight from running stats
__global__ __launch_bounds__(256)
void bn_eval_coeffs_kernel(const float* __restrict__ weight,
                           const float* __restrict__ bias,
                           const float* __restrict__ running_mean,
                           const float* __restrict__ running_var,
                           float* __restrict__ save_mean,
                           float* __restrict__ save_rstd,
                           float* __restrict__ scale_shift,
                           int C, float eps) {
    const int c = blockIdx.x * 256 + threadIdx.x;
    if (c >= C) return;
    const float mean = running_mean[c];
    const float rstd = rsqrtf(running_var[c] + eps);
    save_mean[c] = mean;
    save_rstd[c] = rstd;
    const float sc = weight[c] * rstd;
    scale_shift[c] = sc;
    scale_shift[C + c] = bias[c] - mean * sc;
}

// apply: y = [relu](x*scale + shift [+ res]); RELU/RES are compile-time
// channel-resident thread mapping (as bn_stats): fixed c8 per thread, row
// loop — no per-element integer division, coalesced 16B lanes.
// MASK (default-on since round 2; MGPROTO_BN_MASK=0 disables): emit one
// relu-mask byte per 8-channel chunk so the backward never re-reads y
// (saves 2 bf16 activation passes of backward traffic).
template <bool RELU, bool RES, bool MASK = false>
__global__ __launch_bounds__(256)
void bn_apply_kernel(const short* __restrict__ x,
                     const short* __restrict__ res,
                     const float* __restrict__ scale_shift,
                     short* __restrict__ y,
                     unsigned char* __restrict__ mask, long M, int C) {
    const int tpr = C / 8;
    int c8, rsub, rpi;
    if (tpr >= 256) {
        c8 = (blockIdx.y * 256 + threadIdx.x) * 8;
        if (c8 >= C) return;
        rsub = 0; rpi = 1;
    } else {
        rpi = 256 / tpr;
        rsub = threadIdx.x / tpr;
        c8 = (threadIdx.x % tpr) * 8;
        if (rsub >= rpi) return;
    }
    float sc[8], sh[8];
    #pragma unroll
    for (int i = 0; i < 8; ++i) {
        sc[i] = scale_shift[c8 + i];
        sh[i] = scale_shift[C + c8 + i];
    }
    for (long m = (long)blockIdx.x * rpi + rsub; m < M;
         m += (long)gridDim.x * rpi) {
        const short8 v = *reinterpret_cast<const short8*>(x + m * C + c8);
        short8 r;
        if (RES) r = *reinterpret_cast<const short8*>(res + m * C + c8);
        short8 o;
        unsigned char mb = 0;
        #pragma unroll
        for (int i = 0; i < 8; ++i) {
            float f = fmaf(bf2f(v[i]), sc[i], sh[i]);
            if (RES) f += bf2f(r[i]);
            if (RELU) f = fmaxf(f, 0.f);
            o[i] = f2bf(f);
            // mask from the ROUNDED value: bit-identical to the y>0 test
            // the non-mask backward performs on the stored bf16 y
            if (MASK && RELU && bf2f(o[i]) > 0.f) mb |= (1u << i);
        }
        *reinterpret_cast<short8*>(y + m * C + c8) = o;
        if (MASK) mask[m * (C / 8) + (c8 >> 3)] = mb;
    }
}

// ---------------------------------------------------------------------------
// backward
// ---------------------------------------------------------------------------

// reduce: sum(dy_eff), sum(dy_eff * xhat) per channel; dy_eff = dy * (y>0)
// when the forward fused a ReLU
template <bool RELU, bool MASK = false>
__global__ __launch_bounds__(256)
void bn_bwd_reduce_kernel(const short* __restrict__ dy,
                          const short* __restrict__ y,   // or mask if MASK
                          const short* __restrict__ x,
                          const float* __restrict__ save_mean,
                          const float* __restrict__ save_rstd,
                          float* __restrict__ partials,  // [nblocks, 2C]
                          long M, int C) {
    const unsigned char* __restrict__ msk =
        reinterpret_cast<const unsigned char*>(y);
    __shared__ __attribute__((aligned(16))) float lds[256 * 17];
    const int tpr = C / 8;
    float sd[8] = {0}, sx[8] = {0};
    if (tpr >= 256) {
        const int c8 = (blockIdx.y * 256 + threadIdx.x) * 8;
        float* partial = partials + ((long)blockIdx.y * gridDim.x
                                     + blockIdx.x) * 2 * C;
        if (c8 < C) {
            float mean[8], rstd[8];
            #pragma unroll
            for (int i = 0; i < 8; ++i) {
                mean[i] = save_mean[c8 + i];
                rstd[i] = save_rstd[c8 + i];
            }
            // 2-deep unroll (4-6 loads in flight) — same MLP fix as the
            // small-C path; this branch serves C>=2048 (DenseNet stages)
            const long st = gridDim.x;
            long m = blockIdx.x;
            for (; m + st < M; m += 2 * st) {
                short8 g[2], xv[2], yv[2];
                unsigned char mb[2] = {0xff, 0xff};
                #pragma unroll
                for (int u = 0; u < 2; ++u) {
                    const long mm = m + u * st;
                    g[u] = *reinterpret_cast<const short8*>(dy + mm * C + c8);
                    xv[u] = *reinterpret_cast<const short8*>(x + mm * C + c8);
                    if (RELU && MASK) mb[u] = msk[mm * (C / 8) + (c8 >> 3)];
                    else if (RELU)
                        yv[u] = *reinterpret_cast<const short8*>(y + mm * C + c8);
                }
                #pragma unroll
                for (int u = 0; u < 2; ++u)
                    #pragma unroll
                    for (int i = 0; i < 8; ++i) {
                        float gf = bf2f(g[u][i]);
                        if (RELU && MASK) { if (!((mb[u] >> i) & 1)) gf = 0.f; }
                        else if (RELU && bf2f(yv[u][i]) <= 0.f) gf = 0.f;
                        const float xhat = (bf2f(xv[u][i]) - mean[i]) * rstd[i];
                        sd[i] += gf; sx[i] += gf * xhat;
                    }
            }
            for (; m < M; m += st) {
                const short8 g = *reinterpret_cast<const short8*>(dy + m * C + c8);
                const short8 xv = *reinterpret_cast<const short8*>(x + m * C + c8);
                short8 yv;
                unsigned char mb = 0xff;
                if (RELU && MASK) mb = msk[m * (C / 8) + (c8 >> 3)];
                else if (RELU) yv = *reinterpret_cast<const short8*>(y + m * C + c8);
                #pragma unroll
                for (int i = 0; i < 8; ++i) {
                    float gf = bf2f(g[i]);
                    if (RELU && MASK) { if (!((mb >> i) & 1)) gf = 0.f; }
                    else if (RELU && bf2f(yv[i]) <= 0.f) gf = 0.f;
                    const float xhat = (bf2f(xv[i]) - mean[i]) * rstd[i];
                    sd[i] += gf; sx[i] += gf * xhat;
                }
            }
            #pragma unroll
            for (int i = 0; i < 8; ++i) {
                partial[c8 + i] = sd[i];
                partial[C + c8 + i] = sx[i];
            }
        }
        return;
    }
    const int rpi = 256 / tpr;
    const int rsub = threadIdx.x / tpr;
    const int c8 = (threadIdx.x % tpr) * 8;
    if (rsub < rpi) {
        float mean[8], rstd[8];
        #pragma unroll
        for (int i = 0; i < 8; ++i) {
            mean[i] = save_mean[c8 + i];
            rstd[i] = save_rstd[c8 + i];
        }
        // 2-deep unroll: 4-6 independent loads in flight (see bn_stats)
        const long st = (long)gridDim.x * rpi;
        long m = (long)blockIdx.x * rpi + rsub;
        for (; m + st < M; m += 2 * st) {
            short8 g[2], xv[2], yv[2];
            unsigned char mb[2] = {0xff, 0xff};
            #pragma unroll
            for (int u = 0; u < 2; ++u) {
                const long mm = m + u * st;
                g[u] = *reinterpret_cast<const short8*>(dy + mm * C + c8);
                xv[u] = *reinterpret_cast<const short8*>(x + mm * C + c8);
                if (RELU && MASK) mb[u] = msk[mm * (C / 8) + (c8 >> 3)];
                else if (RELU)
                    yv[u] = *reinterpret_cast<const short8*>(y + mm * C + c8);
            }
            #pragma unroll
            for (int u = 0; u < 2; ++u)
                #pragma unroll
                for (int i = 0; i < 8; ++i) {
                    float gf = bf2f(g[u][i]);
                    if (RELU && MASK) { if (!((mb[u] >> i) & 1)) gf = 0.f; }
                    else if (RELU && bf2f(yv[u][i]) <= 0.f) gf = 0.f;
                    const float xhat = (bf2f(xv[u][i]) - mean[i]) * rstd[i];
                    sd[i] += gf; sx[i] += gf * xhat;
                }
        }
        for (; m < M; m += st) {
            const short8 g = *reinterpret_cast<const short8*>(dy + m * C + c8);
            const short8 xv = *reinterpret_cast<const short8*>(x + m * C + c8);
            short8 yv;
            unsigned char mb = 0xff;
            if (RELU && MASK) mb = msk[m * (C / 8) + (c8 >> 3)];
            else if (RELU) yv = *reinterpret_cast<const short8*>(y + m * C + c8);
            #pragma unroll
            for (int i = 0; i < 8; ++i) {
                float gf = bf2f(g[i]);
                if (RELU && MASK) { if (!((mb >> i) & 1)) gf = 0.f; }
                else if (RELU && bf2f(yv[i]) <= 0.f) gf = 0.f;
                const float xhat = (bf2f(xv[i]) - mean[i]) * rstd[i];
                sd[i] += gf; sx[i] += gf * xhat;
            }
        }
    }
    bn_block_partial(sd, sx, lds, tpr, partials + (long)blockIdx.x * 2 * C, C);
}

// apply: dx = gamma*rstd * (dy_eff - sum_dy/M - xhat*sum_dyxhat/M)
//        d_res = dy_eff (residual branch grad) when RES
template <bool RELU, bool RES, bool MASK = false>
__global__ __launch_bounds__(256)
void bn_bwd_apply_kernel(const short* __restrict__ dy,
                         const short* __restrict__ y,   // or mask if MASK
                         const short* __restrict__ x,
                         const float* __restrict__ save_mean,
                         const float* __restrict__ save_rstd,
                         const float* __restrict__ weight,
                         const float* __restrict__ sums,
                         short* __restrict__ dx,
                         short* __restrict__ dres,
                         long M, int C) {
    const unsigned char* __restrict__ msk =
        reinterpret_cast<const unsigned char*>(y);
    const float invM = 1.f / (float)M;
    const int tpr = C / 8;
    int c8, rsub, rpi;
    if (tpr >= 256) {
        c8 = (blockIdx.y * 256 + threadIdx.x) * 8;
        if (c8 >= C) return;
        rsub = 0; rpi = 1;
    } else {
        rpi = 256 / tpr;
        rsub = threadIdx.x / tpr;
        c8 = (threadIdx.x % tpr) * 8;
        if (rsub >= rpi) return;
    }
    float mean[8], rstd[8], gw[8], sdy[8], sxh[8];
    #pragma unroll
    for (int i = 0; i < 8; ++i) {
        mean[i] = save_mean[c8 + i];
        rstd[i] = save_rstd[c8 + i];
        gw[i] = weight[c8 + i] * rstd[i];
        sdy[i] = sums[c8 + i] * invM;
        sxh[i] = sums[C + c8 + i] * invM;
    }
    for (long m = (long)blockIdx.x * rpi + rsub; m < M;
         m += (long)gridDim.x * rpi) {
        const short8 g = *reinterpret_cast<const short8*>(dy + m * C + c8);
        const short8 xv = *reinterpret_cast<const short8*>(x + m * C + c8);
        short8 yv;
        unsigned char mb = 0xff;
        if (RELU && MASK) mb = msk[m * (C / 8) + (c8 >> 3)];
        else if (RELU) yv = *reinterpret_cast<const short8*>(y + m * C + c8);
        short8 odx, odr;
        #pragma unroll
        for (int i = 0; i < 8; ++i) {
            float gf = bf2f(g[i]);
            if (RELU && MASK) { if (!((mb >> i) & 1)) gf = 0.f; }
            else if (RELU && bf2f(yv[i]) <= 0.f) gf = 0.f;
            if (RES) odr[i] = f2bf(gf);
            const float xhat = (bf2f(xv[i]) - mean[i]) * rstd[i];
            odx[i] = f2bf(gw[i] * (gf - sdy[i] - xhat * sxh[i]));
        }
        *reinterpret_cast<short8*>(dx + m * C + c8) = odx;
        if (RES) *reinterpret_cast<short8*>(dres + m * C + c8) = odr;
    }
}

// eval-mode backward (no batch-stat dependency):
// dx = gamma*rstd*dy_eff
template <bool RELU, bool RES, bool MASK = false>
__global__ __launch_bounds__(256)
void bn_bwd_eval_kernel(const short* __restrict__ dy,
                        const short* __restrict__ y,   // or mask if MASK
                        const float* __restrict__ save_rstd,
                        const float* __restrict__ weight,
                        short* __restrict__ dx,
                        short* __restrict__ dres,
                        long M, int C) {
    const unsigned char* __restrict__ msk =
        reinterpret_cast<const unsigned char*>(y);
    const int tpr = C / 8;
    int c8, rsub, rpi;
    if (tpr >= 256) {
        c8 = (blockIdx.y * 256 + threadIdx.x) * 8;
        if (c8 >= C) return;
        rsub = 0; rpi = 1;
    } else {
        rpi = 256 / tpr;
        rsub = threadIdx.x / tpr;
        c8 = (threadIdx.x % tpr) * 8;
        if (rsub >= rpi) return;
    }
    float gw[8];
    #pragma unroll
    for (int i = 0; i < 8; ++i)
        gw[i] = weight[c8 + i] * save_rstd[c8 + i];
    for (long m = (long)blockIdx.x * rpi + rsub; m < M;
         m += (long)gridDim.x * rpi) {
        const short8 g = *reinterpret_cast<const short8*>(dy + m * C + c8);
        short8 yv;
        unsigned char mb = 0xff;
        if (RELU && MASK) mb = msk[m * (C / 8) + (c8 >> 3)];
        else if (RELU) yv = *reinterpret_cast<const short8*>(y + m * C + c8);
        short8 odx, odr;
        #pragma unroll
        for (int i = 0; i < 8; ++i) {
            float gf = bf2f(g[i]);
            if (RELU && MASK) { if (!((mb >> i) & 1)) gf = 0.f; }
            else if (RELU && bf2f(yv[i]) <= 0.f) gf = 0.f;
            if (RES) odr[i] = f2bf(gf);
            odx[i] = f2bf(gw[i] * gf);
        }
        *reinterpret_cast<short8*>(dx + m * C + c8) = odx;
        if (RES) *reinterpret_cast<short8*>(dres + m * C + c8) = odr;
    }
}

// ---------------------------------------------------------------------------
// host wrappers
// ---------------------------------------------------------------------------

static inline int cdiv(long a, long b) { return (int)((a + b - 1) / b); }

std::vector<torch::Tensor> bn_fwd(torch::Tensor x, torch::Tensor weight,
                                  torch::Tensor bias,
                                  torch::Tensor running_mean,
                                  torch::Tensor running_var,
                                  bool training, double momentum, double eps,
                                  bool relu,
                                  c10::optional<torch::Tensor> residual,
                                  bool want_mask) {
    CHECK_BN(x);
    TORCH_CHECK(x.dtype() == torch::kBFloat16, "bn_fwd: bf16 only");
    const long M = x.size(0);
    const int C = x.size(1);
    TORCH_CHECK(C % 8 == 0, "C must be a multiple of 8");
    auto stream = at::hip::getCurrentHIPStream();
    auto fopt = x.options().dtype(torch::kFloat32);
    auto save_mean = torch::empty({C}, fopt);
    auto save_rstd = torch::empty({C}, fopt);
    auto scale_shift = torch::empty({2 * C}, fopt);
    auto y = torch::empty_like(x);

    const int tpr = C / 8;
    const int grid_y = tpr >= 256 ? cdiv(tpr, 256) : 1;
    auto reduce_grid = [&]() {
        if (tpr >= 256)
            return dim3(std::max(1, std::min((int)M, 64)), grid_y);
        const int rpi = 256 / tpr;
        return dim3((int)std::max<long>(1, std::min<long>(cdiv(M, rpi), 1024)), 1);
    };
    if (training) {
        const dim3 rgrid = reduce_grid();
        const int nb = rgrid.x * rgrid.y;
        // zero-filled for the channel-split path (each block writes only
        // its channel slice of its partial row)
        auto partials = (tpr >= 256)
            ? torch::zeros({nb, 2 * C}, fopt)
            : torch::empty({nb, 2 * C}, fopt);
        auto sums = torch::empty({2 * C}, fopt);
        hipLaunchKernelGGL(bn_stats_kernel, rgrid, dim3(256),
                           0, stream, (const short*)x.data_ptr(),
                           partials.data_ptr<float>(), M, C);
        launch_partial_sum(partials.data_ptr<float>(),
                           sums.data_ptr<float>(), nb, 2 * C, stream);
        hipLaunchKernelGGL(bn_finalize_kernel, dim3(cdiv(C, 256)), dim3(256),
                           0, stream, sums.data_ptr<float>(),
                           weight.data_ptr<float>(), bias.data_ptr<float>(),
                           running_mean.data_ptr<float>(),
                           running_var.data_ptr<float>(),
                           save_mean.data_ptr<float>(),
                           save_rstd.data_ptr<float>(),
                           scale_shift.data_ptr<float>(), M, C,
                           (float)momentum, (float)eps, 1);
    } else {
        hipLaunchKernelGGL(bn_eval_coeffs_kernel, dim3(cdiv(C, 256)), dim3(256),
                           0, stream, weight.data_ptr<float>(),
                           bias.data_ptr<float>(),
                           running_mean.data_ptr<float>(),
                           running_var.data_ptr<float>(),
                           save_mean.data_ptr<float>(),
                           save_rstd.data_ptr<float>(),
                           scale_shift.data_ptr<float>(), C, (float)eps);
    }

    const dim3 agrid = (tpr >= 256)
        ? dim3(std::max(1, std::min((int)M, 1024 / grid_y)), grid_y)
        : dim3((int)std::max<long>(1, std::min<long>(cdiv(M, 256 / tpr), 1024)), 1);
    const short* res_ptr = nullptr;
    if (residual.has_value()) {
        CHECK_BN(residual.value());
        res_ptr = (const short*)residual.value().data_ptr();
    }
    auto mask = (want_mask && relu)
        ? torch::empty({M, C / 8}, x.options().dtype(torch::kUInt8))
        : torch::empty({0}, x.options().dtype(torch::kUInt8));
    unsigned char* mask_ptr = (want_mask && relu)
        ? mask.data_ptr<unsigned char>() : nullptr;
    #define APPLY(RELU_, RES_, MASK_) \
        hipLaunchKernelGGL((bn_apply_kernel<RELU_, RES_, MASK_>), agrid, \
                           dim3(256), 0, stream, (const short*)x.data_ptr(), \
                           res_ptr, scale_shift.data_ptr<float>(), \
                           (short*)y.data_ptr(), mask_ptr, M, C)
    if (mask_ptr && res_ptr) APPLY(true, true, true);
    else if (mask_ptr) APPLY(true, false, true);
    else if (relu && res_ptr) APPLY(true, true, false);
    else if (relu) APPLY(true, false, false);
    else if (res_ptr) APPLY(false, true, false);
    else APPLY(false, false, false);
    #undef APPLY
    return {y, save_mean, save_rstd, mask};
}

std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, torch::Tensor y,
                                  torch::Tensor x, torch::Tensor weight,
                                  torch::Tensor save_mean,
                                  torch::Tensor save_rstd,
                                  bool training, bool relu, bool has_res,
                                  bool use_mask) {
    CHECK_BN(dy); CHECK_BN(x);
    const long M = x.size(0);
    const int C = x.size(1);
    auto stream = at::hip::getCurrentHIPStream();
    auto fopt = x.options().dtype(torch::kFloat32);
    auto dx = torch::empty_like(x);
    auto dres = has_res ? torch::empty_like(x) : torch::empty({0}, x.options());
    auto sums = torch::zeros({2 * C}, fopt);

    const int tpr = C / 8;
    const int grid_y = tpr >= 256 ? cdiv(tpr, 256) : 1;
    auto reduce_grid = [&]() {
        if (tpr >= 256)
            return dim3(std::max(1, std::min((int)M, 64)), grid_y);
        const int rpi = 256 / tpr;
        return dim3((int)std::max<long>(1, std::min<long>(cdiv(M, rpi), 1024)), 1);
    };
    const dim3 rgrid = reduce_grid();
    const int nb = rgrid.x * rgrid.y;
    auto partials = (tpr >= 256)
        ? torch::zeros({nb, 2 * C}, fopt)
        : torch::empty({nb, 2 * C}, fopt);
    const dim3 agrid = (tpr >= 256)
        ? dim3(std::max(1, std::min((int)M, 1024 / grid_y)), grid_y)
        : dim3((int)std::max<long>(1, std::min<long>(cdiv(M, 256 / tpr), 1024)), 1);

    if (training) {
        #define RED(RELU_, MASK_) \
            hipLaunchKernelGGL((bn_bwd_reduce_kernel<RELU_, MASK_>), \
                               rgrid, dim3(256), 0, stream, \
                               (const short*)dy.data_ptr(), \
                               (const short*)y.data_ptr(), \
                               (const short*)x.data_ptr(), \
                               save_mean.data_ptr<float>(), \
                               save_rstd.data_ptr<float>(), \
                               partials.data_ptr<float>(), M, C)
        if (relu && use_mask) RED(true, true);
        else if (relu) RED(true, false);
        else RED(false, false);
        #undef RED
        launch_partial_sum(partials.data_ptr<float>(),
                           sums.data_ptr<float>(), nb, 2 * C, stream);
        #define BWD_APPLY(RELU_, RES_, MASK_) \
            hipLaunchKernelGGL((bn_bwd_apply_kernel<RELU_, RES_, MASK_>), \
                               agrid, dim3(256), 0, stream, \
                               (const short*)dy.data_ptr(), \
                               (const short*)y.data_ptr(), \
                               (const short*)x.data_ptr(), \
                               save_mean.data_ptr<float>(), \
                               save_rstd.data_ptr<float>(), \
                               weight.data_ptr<float>(), \
                               sums.data_ptr<float>(), \
                               (short*)dx.data_ptr(), \
                               has_res ? (short*)dres.data_ptr() : nullptr, M, C)
        if (relu && use_mask && has_res) BWD_APPLY(true, true, true);
        else if (relu && use_mask) BWD_APPLY(true, false, true);
        else if (relu && has_res) BWD_APPLY(true, true, false);
        else if (relu) BWD_APPLY(true, false, false);
        else if (has_res) BWD_APPLY(false, true, false);
        else BWD_APPLY(false, false, false);
        #undef BWD_APPLY
    } else {
        // eval: dx = gamma*rstd*dy_eff; grads for weight/bias still need the
        // reduce (xhat uses running stats)
        #define RED(RELU_, MASK_) \
            hipLaunchKernelGGL((bn_bwd_reduce_kernel<RELU_, MASK_>), \
                               rgrid, dim3(256), 0, stream, \
                               (const short*)dy.data_ptr(), \
                               (const short*)y.data_ptr(), \
                               (const short*)x.data_ptr(), \
                               save_mean.data_ptr<float>(), \
                               save_rstd.data_ptr<float>(), \
                               partials.data_ptr<float>(), M, C)
        if (relu && use_mask) RED(true, true);
        else if (relu) RED(true, false);
        else RED(false, false);
        #undef RED
        launch_partial_sum(partials.data_ptr<float>(),
                           sums.data_ptr<float>(), nb, 2 * C, stream);
        #define EVAL_APPLY(RELU_, RES_, MASK_) \
            hipLaunchKernelGGL((bn_bwd_eval_kernel<RELU_, RES_, MASK_>), \
                               agrid, dim3(256), 0, stream, \
                               (const short*)dy.data_ptr(), \
                               (const short*)y.data_ptr(), \
                               save_rstd.data_ptr<float>(), \
                               weight.data_ptr<float>(), \
                               (short*)dx.data_ptr(), \
                               has_res ? (short*)dres.data_ptr() : nullptr, M, C)
        if (relu && use_mask && has_res) EVAL_APPLY(true, true, true);
        else if (relu && use_mask) EVAL_APPLY(true, false, true);
        else if (relu && has_res) EVAL_APPLY(true, true, false);
        else if (relu) EVAL_APPLY(true, false, false);
        else if (has_res) EVAL_APPLY(false, true, false);
        else EVAL_APPLY(false, false, false);
        #undef EVAL_APPLY
    }
    // dgamma = sum(dy_eff * xhat), dbeta = sum(dy_eff)
    auto dweight = sums.narrow(0, C, C).clone();
    auto dbias = sums.narrow(0, 0, C).clone();
    return {dx, dweight, dbias, dres};
}
