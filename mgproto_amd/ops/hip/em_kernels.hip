// Batched EM kernels for MI355X (gfx950, CDNA4) — K6/K7 of SURVEY.md §2.2.
//
// The EM update runs over all dirty classes at once: x [G, N, d] memory
// features, K Gaussian components per class.  Same math as the torch path
// (ops/reference.py em_e_step / em_m_step_grads, itself verified against
// the reference's autograd at model.py:303-401):
//
//   e-step   : wlp[n,k] = bias[k] + sum_j x A[k] + x^2 B[k]   (+ log pi)
//              logresp  = wlp - logsumexp_k wlp
//   m-step   : resp'    = (exp(logresp) + alpha) / (1 + alpha K)
//              rx[k,j]  = sum_n resp' x[n,j],   rsum[k] = sum_n resp'
//              grad     = -(rx - rsum mu) / sig^2 / N
//                         - lamda (4/S) sum_b w_kb (mu_k - mu_b),
//              w_kb = exp(-||mu_k - mu_b||^2), S = max(K^2-K, 1)
//
// The O(G N K d) work runs here; the O(G K d) parameter prep (A/B/bias
// from means/covs/pi) stays in torch on device.  Shapes: d <= 128,
// K <= 32 (the dispatch falls back to the torch path outside this
// envelope).  All kernels are launch-capturable (no host syncs) and run
// on the current HIP stream, so the EM side-stream overlap and whole-step
// hipGraph capture keep working.
//
// Gated opt-in (MGPROTO_HIP_EM=1) until GPU-validated; the default EM path
// remains the rocBLAS baddbmm form.

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <ATen/hip/HIPContext.h>

#define CHECK_EM(x) TORCH_CHECK(x.is_cuda() && x.is_contiguous(), #x " must be contiguous on device")

static inline int ceil_div_em(int a, int b) { return (a + b - 1) / b; }

__device__ inline float wave_sum(float v) {
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        v += __shfl_down(v, off, 64);
    return v;
}

// ---------------------------------------------------------------------------
// E-step: one wavefront per sample row; lane j covers dims j, j+64.
// grid (ceil(N/4), G), block 256 = 4 waves.
// ---------------------------------------------------------------------------

template <int KMAX>
__global__ __launch_bounds__(256)
void em_estep_kernel(const float* __restrict__ x,      // [G, N, d]
                     const float* __restrict__ A,      // [G, K, d]
                     const float* __restrict__ B,      // [G, K, d]
                     const float* __restrict__ bias,   // [G, K] (incl log pi)
                     float* __restrict__ wlp,          // [G, N, K]
                     float* __restrict__ logresp,      // [G, N, K]
                     int N, int K, int d) {
    const int g = blockIdx.y;
    const int tid = threadIdx.x;
    const int wave = tid >> 6, lane = tid & 63;
    const int n = blockIdx.x * 4 + wave;

    // stage this class's A/B/bias in LDS (K*d <= 32*128 = 4096 floats each)
    extern __shared__ float lds[];
    float* a_s = lds;                    // [K*d]
    float* b_s = a_s + K * d;            // [K*d]
    float* c_s = b_s + K * d;            // [K]
    for (int i = tid; i < K * d; i += 256) {
        a_s[i] = A[((size_t)g * K) * d + i];
        b_s[i] = B[((size_t)g * K) * d + i];
    }
    for (int i = tid; i < K; i += 256)
        c_s[i] = bias[(size_t)g * K + i];
    __syncthreads();

    if (n >= N) return;

    // own dims into registers (d <= 128 -> at most 2 per lane)
    const float* xrow = x + ((size_t)g * N + n) * d;
    float xr0 = 0.f, xr1 = 0.f;
    if (lane < d) xr0 = xrow[lane];
    if (lane + 64 < d) xr1 = xrow[lane + 64];
    const float x20 = xr0 * xr0, x21 = xr1 * xr1;

    // early-exit loops: measured better than full predication here
    // (52 vs 110 VGPRs; acc[] slots beyond K die, no spills, 8 waves/SIMD)
    float acc[KMAX];
    #pragma unroll
    for (int k = 0; k < KMAX; ++k) {
        if (k >= K) break;
        float p = 0.f;
        if (lane < d)
            p += xr0 * a_s[k * d + lane] + x20 * b_s[k * d + lane];
        if (lane + 64 < d)
            p += xr1 * a_s[k * d + lane + 64] + x21 * b_s[k * d + lane + 64];
        p = wave_sum(p);
        acc[k] = c_s[k] + p;             // valid on lane 0
    }

    if (lane != 0) return;
    float m = -INFINITY;
    #pragma unroll
    for (int k = 0; k < KMAX; ++k) {
        if (k >= K) break;
        m = fmaxf(m, acc[k]);
    }
    float s = 0.f;
    #pragma unroll
    for (int k = 0; k < KMAX; ++k) {
        if (k >= K) break;
        s += __expf(acc[k] - m);
    }
    const float lse = m + __logf(s);
    float* wrow = wlp + ((size_t)g * N + n) * K;
    float* rrow = logresp + ((size_t)g * N + n) * K;
    #pragma unroll
    for (int k = 0; k < KMAX; ++k) {
        if (k >= K) break;
        wrow[k] = acc[k];
        rrow[k] = acc[k] - lse;
    }
}

// ---------------------------------------------------------------------------
// M-step accumulation: one block per (g, k); thread j owns dim j.
// rx[g,k,j] = sum_n resp' x[n,j]; rsum[g,k] = sum_n resp'.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(128)
void em_mstep_accum_kernel(const float* __restrict__ x,        // [G, N, d]
                           const float* __restrict__ logresp,  // [G, N, K]
                           float alpha,
                           float* __restrict__ rx,             // [G, K, d]
                           float* __restrict__ rsum,           // [G, K]
                           int N, int K, int d) {
    const int g = blockIdx.x / K;
    const int k = blockIdx.x % K;
    const int j = threadIdx.x;

    float acc = 0.f, accr = 0.f;
    const float* xg = x + (size_t)g * N * d;
    const float* lrow = logresp + (size_t)g * N * K;
    for (int n = 0; n < N; ++n) {
        // same smoothing denominator as the oracle: the ACTUAL row sum
        // (sum_k exp(logresp) is 1 only up to fp error)
        float den = 0.f;
        for (int kk = 0; kk < K; ++kk)
            den += __expf(lrow[(size_t)n * K + kk]) + alpha;
        const float r = (__expf(lrow[(size_t)n * K + k]) + alpha) / den;
        if (j < d) acc += r * xg[(size_t)n * d + j];
        accr += r;
    }
    if (j < d) rx[((size_t)g * K + k) * d + j] = acc;
    if (j == 0) rsum[(size_t)g * K + k] = accr;
}

// ---------------------------------------------------------------------------
// Combine: grad = grad_nll + lamda * grad_div; pi_unnorm = rsum + eps.
// One block per class g.
// ---------------------------------------------------------------------------

template <int KMAX>
__global__ __launch_bounds__(256)
void em_combine_kernel(const float* __restrict__ rx,      // [G, K, d]
                       const float* __restrict__ rsum,    // [G, K]
                       const float* __restrict__ means,   // [G, K, d]
                       const float* __restrict__ covs,    // [G, K, d]
                       float lamda, float eps, int N, int K, int d,
                       float* __restrict__ grad,          // [G, K, d]
                       float* __restrict__ pi_unnorm) {   // [G, K]
    const int g = blockIdx.x;
    const int tid = threadIdx.x;

    extern __shared__ float lds[];
    float* mu_s = lds;                   // [K*d]
    float* w_s = mu_s + K * d;           // [K*K]
    for (int i = tid; i < K * d; i += 256)
        mu_s[i] = means[((size_t)g * K) * d + i];
    __syncthreads();

    // pairwise repulsion weights w_ib = exp(-||mu_i - mu_b||^2), diag 0
    for (int p = tid; p < K * K; p += 256) {
        const int i = p / K, b = p % K;
        float dist = 0.f;
        for (int j = 0; j < d; ++j) {
            const float diff = mu_s[i * d + j] - mu_s[b * d + j];
            dist += diff * diff;
        }
        w_s[p] = (i == b) ? 0.f : __expf(-dist);
    }
    __syncthreads();

    const float S = (float)max(K * K - K, 1);
    const float div_scale = -lamda * 4.0f / S;
    for (int cell = tid; cell < K * d; cell += 256) {
        const int k = cell / d, j = cell % d;
        const float mu = mu_s[cell];
        const float sig = covs[((size_t)g * K) * d + cell] + eps;
        const float iv = 1.0f / (sig * sig);
        const float rs = rsum[(size_t)g * K + k];
        const float gn = -(rx[((size_t)g * K) * d + cell] - rs * mu)
                         * iv / (float)N;
        float gd = 0.f;
        #pragma unroll
        for (int b = 0; b < KMAX; ++b) {
            if (b >= K) break;
            gd += w_s[k * K + b] * (mu - mu_s[b * d + j]);
        }
        grad[((size_t)g * K) * d + cell] = gn + div_scale * gd;
    }
    for (int k = tid; k < K; k += 256)
        pi_unnorm[(size_t)g * K + k] = rsum[(size_t)g * K + k] + eps;
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

std::vector<torch::Tensor> em_estep(torch::Tensor x, torch::Tensor A,
                                    torch::Tensor B, torch::Tensor bias) {
    CHECK_EM(x); CHECK_EM(A); CHECK_EM(B); CHECK_EM(bias);
    TORCH_CHECK(x.dtype() == torch::kFloat32, "em_estep: fp32 only");
    const int G = x.size(0), N = x.size(1), d = x.size(2);
    const int K = A.size(1);
    TORCH_CHECK(A.size(0) == G && A.size(2) == d, "A must be [G, K, d]");
    TORCH_CHECK(d <= 128 && K <= 32, "envelope: d <= 128, K <= 32");
    auto wlp = torch::empty({G, N, K}, x.options());
    auto logresp = torch::empty({G, N, K}, x.options());
    auto stream = at::hip::getCurrentHIPStream();
    const dim3 grid(ceil_div_em(N, 4), G);
    const size_t lds = (2 * K * d + K) * sizeof(float);
    hipLaunchKernelGGL((em_estep_kernel<32>), grid, dim3(256), lds, stream,
                       x.data_ptr<float>(), A.data_ptr<float>(),
                       B.data_ptr<float>(), bias.data_ptr<float>(),
                       wlp.data_ptr<float>(), logresp.data_ptr<float>(),
                       N, K, d);
    return {wlp, logresp};
}

std::vector<torch::Tensor> em_mstep(torch::Tensor x, torch::Tensor logresp,
                                    torch::Tensor means, torch::Tensor covs,
                                    double alpha, double lamda, double eps) {
    CHECK_EM(x); CHECK_EM(logresp); CHECK_EM(means); CHECK_EM(covs);
    TORCH_CHECK(x.dtype() == torch::kFloat32, "em_mstep: fp32 only");
    const int G = x.size(0), N = x.size(1), d = x.size(2);
    const int K = logresp.size(2);
    TORCH_CHECK(means.size(1) == K && means.size(2) == d,
                "means must be [G, K, d]");
    TORCH_CHECK(d <= 128 && K <= 32, "envelope: d <= 128, K <= 32");
    auto rx = torch::empty({G, K, d}, x.options());
    auto rsum = torch::empty({G, K}, x.options());
    auto grad = torch::empty({G, K, d}, x.options());
    auto pi_unnorm = torch::empty({G, K}, x.options());
    auto stream = at::hip::getCurrentHIPStream();
    hipLaunchKernelGGL(em_mstep_accum_kernel, dim3(G * K), dim3(128), 0,
                       stream, x.data_ptr<float>(), logresp.data_ptr<float>(),
                       (float)alpha, rx.data_ptr<float>(),
                       rsum.data_ptr<float>(), N, K, d);
    const size_t lds = (K * d + K * K) * sizeof(float);
    hipLaunchKernelGGL((em_combine_kernel<32>), dim3(G), dim3(256), lds,
                       stream, rx.data_ptr<float>(), rsum.data_ptr<float>(),
                       means.data_ptr<float>(), covs.data_ptr<float>(),
                       (float)lamda, (float)eps, N, K, d,
                       grad.data_ptr<float>(), pi_unnorm.data_ptr<float>());
    return {grad, pi_unnorm};
}
