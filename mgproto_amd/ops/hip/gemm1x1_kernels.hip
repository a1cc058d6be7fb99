// Hand-written bf16 MFMA GEMM for the stride-1 1x1 convs (round-3 lever,
// docs/DESIGN_ROUND3_CONV.md): y[M,N] = x[M,K] @ W[N,K]^T (+bias), bf16
// in / fp32 accumulate / bf16 out — exactly the GEMM a 1x1 conv on an
// NHWC tensor is. The library avenues are measured closed (MIOpen SEARCH
// neutral, TunableOp-hipBLASLt slower, per-shape hybrid 0.04 ms); this
// kernel exists so the EPILOGUE can fuse (BN stats first): bn_partial
// emits one per-channel [2N] partial row per block, merged by the
// existing bn stage-2 — the separate bn_stats pass over y disappears.
//
// v1 is correctness-first per the CDNA4 guide's canonical GEMM anatomy:
// single-buffered LDS staging, one ds_read_b128 per fragment,
// v_mfma_f32_16x16x32_bf16 (C/D: col=lane&15, row=(lane>>4)*4+reg;
// A/B: k=(lane>>4)*8+idx — same lane convention as the proven fp32 gmm
// kernels). Double-buffering / global_load_lds / 8-phase interleave are
// round-3 work once parity + a baseline number exist. Opt-in:
// MGPROTO_GEMM1X1_HIP=1 (tests gated the same way).

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <ATen/hip/HIPContext.h>

using short8 = __attribute__((ext_vector_type(8))) short;   // 8 bf16
using f32x4g = __attribute__((ext_vector_type(4))) float;

static __device__ __forceinline__ float bf2f_g(short s) {
    union { float f; unsigned u; } c;
    c.u = ((unsigned)(unsigned short)s) << 16;
    return c.f;
}
static __device__ __forceinline__ short f2bf_g(float f) {
    union { float f; unsigned u; } c; c.f = f;
    unsigned r = 0x7FFF + ((c.u >> 16) & 1);
    return (short)((c.u + r) >> 16);
}

// BM x BN block tile, BK=64 K-tile, 256 threads = 4 waves in a 2x2 wave
// grid; KS = BK + 8 shorts keeps short8 row reads 16B-aligned and the
// 16-lane b128 read phases bank-clean (stride 144 B).
// MODE 0: plain staged; 1: register double-buffer (K>=1024 winner);
// 2: global_load_lds direct staging into an XOR-swizzled LINEAR layout
// (the LDS-DMA writes wave-uniform-base + lane*16, so padding is
// impossible — the swizzle keeps ds_read_b128 phases bank-clean and the
// PER-LANE GLOBAL address carries it, per the CDNA4 guide's caveat).
template <int BM, int BN, int MODE>
__global__ __launch_bounds__(256)
void gemm1x1_fwd_kernel(const short* __restrict__ x,   // [M, K] bf16
                        const short* __restrict__ w,   // [N, K] bf16
                        const float* __restrict__ bias,// [N] or nullptr
                        short* __restrict__ y,         // [M, N] bf16
                        float* __restrict__ bn_partial,// [nblk, 2N] or null
                        int M, int K, int N) {
    constexpr int BK = 64;
    constexpr bool GLDS = MODE == 2;
    constexpr bool DBUF = MODE == 1;
    constexpr int KS = GLDS ? BK : BK + 8;   // DMA needs the linear layout
    constexpr int WM = BM / 2;
    constexpr int WN = BN / 2;
    constexpr int FM = WM / 16;
    constexpr int FN = WN / 16;

    __shared__ __attribute__((aligned(16))) short lds[(BM + BN) * KS];
    short* As = lds;                 // [BM][KS]
    short* Bs = lds + BM * KS;       // [BN][KS]

    const int n0 = blockIdx.x * BM;  // rows (M)
    const int p0 = blockIdx.y * BN;  // cols (N)
    const int tid = threadIdx.x;
    const int lane = tid & 63;
    const int wave = tid >> 6;
    const int wr = (wave >> 1) * WM;
    const int wc = (wave & 1) * WN;
    const int lrow = lane & 15;
    const int kk = lane >> 4;        // 0..3 -> k-slot of 8

    f32x4g acc[FM][FN];
    #pragma unroll
    for (int i = 0; i < FM; ++i)
        #pragma unroll
        for (int j = 0; j < FN; ++j)
            acc[i][j] = (f32x4g){0.f, 0.f, 0.f, 0.f};

    // Register-buffered double buffering (v1.5): the NEXT K-tile's global
    // loads issue before this tile's MFMAs, so HBM latency hides behind
    // compute; the registers flush to LDS between tiles. No extra LDS.
    constexpr int VEC = BK / 8;                    // short8 per row
    constexpr int A_P = BM * VEC / 256;            // pieces per thread
    constexpr int B_P = BN * VEC / 256;

    short8 ra[A_P], rb[B_P];
    auto load_tile = [&](int k0) {
        #pragma unroll
        for (int p = 0; p < A_P; ++p) {
            const int t = tid + p * 256;
            const int r = t / VEC, c8 = t % VEC;
            const int m = min(n0 + r, M - 1);
            ra[p] = *reinterpret_cast<const short8*>(
                x + (long)m * K + k0 + c8 * 8);
        }
        #pragma unroll
        for (int p = 0; p < B_P; ++p) {
            const int t = tid + p * 256;
            const int r = t / VEC, c8 = t % VEC;
            const int n = min(p0 + r, N - 1);
            rb[p] = *reinterpret_cast<const short8*>(
                w + (long)n * K + k0 + c8 * 8);
        }
    };
    auto flush_tile = [&]() {
        #pragma unroll
        for (int p = 0; p < A_P; ++p) {
            const int t = tid + p * 256;
            *reinterpret_cast<short8*>(As + (t / VEC) * KS + (t % VEC) * 8)
                = ra[p];
        }
        #pragma unroll
        for (int p = 0; p < B_P; ++p) {
            const int t = tid + p * 256;
            *reinterpret_cast<short8*>(Bs + (t / VEC) * KS + (t % VEC) * 8)
                = rb[p];
        }
    };

    // GLDS: each wave issues 1KB direct global->LDS copies; lane i of a
    // call covers LDS piece base+i, and its GLOBAL address is pre-swizzled
    // (chunk c8 = slot ^ (r & 7)) so reads stay conflict-free without pads
    auto glds_tile = [&](int k0) {
        constexpr int A_CALLS = BM / 8;            // 64 16B-pieces per call
        constexpr int B_CALLS = BN / 8;
        constexpr int A_PW = A_CALLS / 4;          // calls per wave
        constexpr int B_PW = B_CALLS / 4;
        #pragma unroll
        for (int q = 0; q < A_PW; ++q) {
            const int call = wave * A_PW + q;
            const int piece = call * 64 + lane;
            const int r = piece >> 3;
            const int c8 = (piece & 7) ^ (r & 7);
            const long m = min(n0 + r, M - 1);
            const short* g = x + (long)m * K + k0 + c8 * 8;
            auto* g1 = (const __attribute__((address_space(1))) void*)(
                reinterpret_cast<uintptr_t>(g));
            auto* l3 = (__attribute__((address_space(3))) void*)(
                reinterpret_cast<uintptr_t>(As + call * 512));
            __builtin_amdgcn_global_load_lds(g1, l3, 16, 0, 0);
        }
        #pragma unroll
        for (int q = 0; q < B_PW; ++q) {
            const int call = wave * B_PW + q;
            const int piece = call * 64 + lane;
            const int r = piece >> 3;
            const int c8 = (piece & 7) ^ (r & 7);
            const int n = min(p0 + r, N - 1);
            const short* g = w + (long)n * K + k0 + c8 * 8;
            auto* g1 = (const __attribute__((address_space(1))) void*)(
                reinterpret_cast<uintptr_t>(g));
            auto* l3 = (__attribute__((address_space(3))) void*)(
                reinterpret_cast<uintptr_t>(Bs + call * 512));
            __builtin_amdgcn_global_load_lds(g1, l3, 16, 0, 0);
        }
    };

    // DBUF pays only when there are many K-tiles to overlap (measured:
    // wins at K>=1024, loses below — the spare registers cost occupancy)
    if constexpr (GLDS) {
        glds_tile(0);
    } else {
        load_tile(0);
        flush_tile();
    }
    __syncthreads();
    for (int k0 = 0; k0 < K; k0 += BK) {
        const bool more = k0 + BK < K;
        if (DBUF && more) load_tile(k0 + BK);      // overlaps the MFMAs below

        #pragma unroll
        for (int ks = 0; ks < BK; ks += 32) {      // two 32-k MFMA steps
            const int qk = (ks >> 3) + kk;         // 16B chunk index 0..7
            short8 afr[FM], bfr[FN];
            #pragma unroll
            for (int i = 0; i < FM; ++i) {
                const int r = wr + i * 16 + lrow;
                const int sl = GLDS ? (qk ^ (r & 7)) * 8 : ks + 8 * kk;
                afr[i] = *reinterpret_cast<const short8*>(As + r * KS + sl);
            }
            #pragma unroll
            for (int j = 0; j < FN; ++j) {
                const int r = wc + j * 16 + lrow;
                const int sl = GLDS ? (qk ^ (r & 7)) * 8 : ks + 8 * kk;
                bfr[j] = *reinterpret_cast<const short8*>(Bs + r * KS + sl);
            }
            #pragma unroll
            for (int i = 0; i < FM; ++i)
                #pragma unroll
                for (int j = 0; j < FN; ++j)
                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        afr[i], bfr[j], acc[i][j], 0, 0, 0);
        }
        __syncthreads();                           // LDS reads done
        if (more) {
            if constexpr (GLDS) {
                glds_tile(k0 + BK);
            } else {
                if (!DBUF) load_tile(k0 + BK);     // plain staged path
                flush_tile();
            }
            __syncthreads();                       // LDS writes visible
        }
    }

    // epilogue: bias, bf16 store, optional per-block BN partials
    float psum[FN][4], psq[FN][4];
    if (bn_partial) {
        #pragma unroll
        for (int j = 0; j < FN; ++j)
            #pragma unroll
            for (int r = 0; r < 4; ++r) { psum[j][r] = 0.f; psq[j][r] = 0.f; }
    }
    #pragma unroll
    for (int i = 0; i < FM; ++i) {
        #pragma unroll
        for (int j = 0; j < FN; ++j) {
            const int col = p0 + wc + j * 16 + lrow;
            const bool colok = col < N;
            const float b = (bias && colok) ? bias[col] : 0.f;
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int row = n0 + wr + i * 16 + (kk * 4 + r);
                if (!colok || row >= M) continue;
                const float v = acc[i][j][r] + b;
                const short o = f2bf_g(v);
                y[(long)row * N + col] = o;
                if (bn_partial) {
                    const float vr = bf2f_g(o);   // stats of the ROUNDED y
                    psum[j][r] += vr; psq[j][r] += vr * vr;
                }
            }
        }
    }
    if (bn_partial) {
        // per-column partials: lane groups (kk, r) hold disjoint row sets
        // of the same column -> combine through LDS (reuse, post-barrier)
        __syncthreads();
        float* red = reinterpret_cast<float*>(lds);   // [BN][2]
        // serialize 32 (wave-row, kk, r) slots: the two wave ROW-halves
        // cover the same columns, so they must not add concurrently
        // (v1 correctness-first; a tree combine is round-3 polish)
        for (int s = 0; s < 32; ++s) {
            const int wrow = s >> 4, skk = (s >> 2) & 3, sr = s & 3;
            if ((wave >> 1) == wrow && kk == skk) {
                #pragma unroll
                for (int j = 0; j < FN; ++j) {
                    const int cl = wc + j * 16 + lrow;   // [0, BN)
                    if (s == 0) {
                        red[cl * 2] = psum[j][sr];
                        red[cl * 2 + 1] = psq[j][sr];
                    } else {
                        red[cl * 2] += psum[j][sr];
                        red[cl * 2 + 1] += psq[j][sr];
                    }
                }
            }
            __syncthreads();
        }
        // one partial row [2N] per (block-row, block-col): layout
        // bn_partial[blk][n] = sum, bn_partial[blk][N + n] = sumsq
        const long blk = (long)blockIdx.x * gridDim.y + blockIdx.y;
        for (int t = tid; t < BN; t += 256) {
            const int col = p0 + t;
            if (col >= N) continue;
            bn_partial[blk * 2 * N + col] = red[t * 2];
            bn_partial[blk * 2 * N + N + col] = red[t * 2 + 1];
        }
    }
}

#define CHECK_G(t) TORCH_CHECK(t.is_cuda() && t.is_contiguous(), #t " must be contiguous on device")

std::vector<torch::Tensor> gemm1x1_fwd(torch::Tensor x, torch::Tensor w,
                                       c10::optional<torch::Tensor> bias,
                                       bool want_bn_partials,
                                       int64_t mode /* -1 auto, 0/1/2 */) {
    CHECK_G(x); CHECK_G(w);
    TORCH_CHECK(x.dtype() == torch::kBFloat16 && w.dtype() == torch::kBFloat16,
                "gemm1x1_fwd: bf16 only");
    const long M = x.size(0);
    const int K = x.size(1), N = w.size(0);
    TORCH_CHECK(w.size(1) == K, "w must be [N, K]");
    TORCH_CHECK(K % 64 == 0, "K must be a multiple of 64");
    TORCH_CHECK(N % 16 == 0, "N must be a multiple of 16");
    auto y = torch::empty({M, N}, x.options());
    auto stream = at::hip::getCurrentHIPStream();
    const float* bias_ptr = nullptr;
    if (bias.has_value()) {
        CHECK_G(bias.value());
        TORCH_CHECK(bias->dtype() == torch::kFloat32, "bias must be fp32");
        bias_ptr = bias->data_ptr<float>();
    }
    // wider M-tile for the tall-skinny shapes, square for big N
    const bool wide = N <= 256;
    const int BM = wide ? 256 : 128;
    const int BN = wide ? 64 : 128;
    dim3 grid((unsigned)((M + BM - 1) / BM), (unsigned)((N + BN - 1) / BN));
    // zeros, not empty: each block writes only ITS column tile of its
    // partial row; the other columns must contribute 0 to the merge
    auto partials = want_bn_partials
        ? torch::zeros({(long)grid.x * grid.y, 2L * N},
                       x.options().dtype(torch::kFloat32))
        : torch::empty({0}, x.options().dtype(torch::kFloat32));
    float* pp = want_bn_partials ? partials.data_ptr<float>() : nullptr;
    // auto = plain staging: the reg-dbuf crossover at K>=1024 did not
    // reproduce across boxes (plain won everywhere on the second box,
    // bench_r02/gemm1x1_hip_v2.log) and glds ~= plain within noise; the
    // modes remain selectable for the round-3 pipeline work
    const int m_ = mode >= 0 ? (int)mode : 0;
    #define LAUNCH_G(BM_, BN_, MD_) \
        hipLaunchKernelGGL((gemm1x1_fwd_kernel<BM_, BN_, MD_>), grid, \
                           dim3(256), 0, stream, (const short*)x.data_ptr(), \
                           (const short*)w.data_ptr(), bias_ptr, \
                           (short*)y.data_ptr(), pp, (int)M, K, N)
    if (wide) {
        if (m_ == 2) LAUNCH_G(256, 64, 2);
        else if (m_ == 1) LAUNCH_G(256, 64, 1);
        else LAUNCH_G(256, 64, 0);
    } else {
        if (m_ == 2) LAUNCH_G(128, 128, 2);
        else if (m_ == 1) LAUNCH_G(128, 128, 1);
        else LAUNCH_G(128, 128, 0);
    }
    #undef LAUNCH_G
    return {y, partials};
}
