// Memory-bank enqueue kernels for MI355X (gfx950) — K5 of SURVEY.md §2.2.
//
// Two pieces, matching the torch path bit-for-bit (ops/reference.py
// enqueue_candidates + utils/memory.py MemoryBank.push, themselves the
// verified rewrite of reference model.py:228-252 / utils/memory.py:32-73):
//
//   enqueue_rows : per sample, the K top-1 patch indices of its GT class's
//                  prototypes, deduplicated (ascending patch order), rows
//                  ordered (class asc, batch asc); duplicates carry the
//                  sentinel label C.  Emits (row index into the packed
//                  feature tensor, label) — the feature gather itself is a
//                  single index_select.
//   bank_push    : class-segregated FIFO ring write of (feature, label)
//                  rows: block c computes each matching row's stable
//                  within-class rank (blocked prefix scan), keeps the
//                  newest `cap` on oversized pushes, writes rows to ring
//                  positions, and advances head/mem_len. Sentinel rows
//                  are discarded (the model's dirty-flag index_fill_
//                  stays a separate single launch).
//
// Deterministic by construction (no atomics; one writer per destination),
// zero host syncs (hipGraph-capturable), current-stream launches.
// Opt-in via MGPROTO_HIP_ENQUEUE=1 until GPU-validated; the default path
// stays the batched torch sort/scan/scatter.

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <ATen/hip/HIPContext.h>

#define CHECK_NQ(x) TORCH_CHECK(x.is_cuda() && x.is_contiguous(), #x " must be contiguous on device")

// ---------------------------------------------------------------------------
// enqueue_rows: ONE block; B <= 1024, K <= 32.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256)
void enqueue_rows_kernel(const int64_t* __restrict__ top1,   // [B, C*K]
                         const int64_t* __restrict__ gt,     // [B]
                         int64_t* __restrict__ rows,         // [B*K]
                         int64_t* __restrict__ lab,          // [B*K]
                         int B, int C, int K, int HW) {
    const int tid = threadIdx.x;
    extern __shared__ int lds_i[];
    int* gts = lds_i;            // [B]
    int* inv = gts + B;          // [B]: inv[slot] = source sample

    for (int b = tid; b < B; b += 256)
        gts[b] = (int)gt[b];
    __syncthreads();

    // stable (class, batch) rank of each sample -> slot
    for (int b = tid; b < B; b += 256) {
        int r = 0;
        for (int b2 = 0; b2 < B; ++b2)
            r += (gts[b2] < gts[b]) || (gts[b2] == gts[b] && b2 < b);
        inv[r] = b;
    }
    __syncthreads();

    for (int row = tid; row < B * K; row += 256) {
        const int slot = row / K, k = row % K;
        const int b = inv[slot];
        const int c = gts[b];
        const int64_t* own = top1 + (size_t)b * C * K + (size_t)c * K;
        // k-th and (k-1)-th smallest of the K values (selection by rank;
        // ties broken by position, which leaves the VALUE sequence equal
        // to a sorted row)
        int64_t vk = 0, vprev = 0;
        for (int i = 0; i < K; ++i) {
            const int64_t v = own[i];
            int r = 0;
            for (int j = 0; j < K; ++j)
                r += (own[j] < v) || (own[j] == v && j < i);
            if (r == k) vk = v;
            if (r == k - 1) vprev = v;
        }
        const bool first = (k == 0) || (vk != vprev);
        rows[row] = (int64_t)b * HW + vk;
        lab[row] = first ? (int64_t)c : (int64_t)C;
    }
}

// ---------------------------------------------------------------------------
// bank_push: one block per class c in [0, C]; c == C is the sentinel class
// (rows discarded). Blocked-partition stable prefix ranks, ring write.
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256)
void bank_push_kernel(const float* __restrict__ feats,    // [M, d]
                      const int64_t* __restrict__ labels, // [M]
                      float* __restrict__ mem,            // [(C+1)*cap, d]
                      int64_t* __restrict__ head,         // [C]
                      int64_t* __restrict__ mem_len,      // [C]
                      int M, int cap, int d) {
    const int c = blockIdx.x;
    const int tid = threadIdx.x;
    __shared__ int cnt_s[257];

    const int chunk = (M + 255) / 256;
    const int lo = tid * chunk, hi = min(lo + chunk, M);
    int cnt = 0;
    for (int r = lo; r < hi; ++r)
        cnt += (labels[r] == c);
    cnt_s[tid + 1] = cnt;
    __syncthreads();
    if (tid == 0) {
        cnt_s[0] = 0;
        for (int t = 0; t < 256; ++t)
            cnt_s[t + 1] += cnt_s[t];          // inclusive -> exclusive base
    }
    __syncthreads();
    const int count = cnt_s[256];
    if (count == 0) return;

    const int h = (int)head[c];
    int rank = cnt_s[tid];
    for (int r = lo; r < hi; ++r) {
        if (labels[r] != c) continue;
        // oversized push: keep only the newest `cap` rows
        if (rank >= count - cap) {
            const int pos = (h + rank) % cap;
            const float* src = feats + (size_t)r * d;
            float* dst = mem + ((size_t)c * cap + pos) * d;
            for (int j = 0; j < d; ++j)
                dst[j] = src[j];
        }
        ++rank;
    }
    __syncthreads();
    if (tid == 0) {
        head[c] = (head[c] + count) % cap;
        mem_len[c] = min((int)mem_len[c] + count, cap);
    }
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------

std::vector<torch::Tensor> enqueue_rows(torch::Tensor top1, torch::Tensor gt,
                                        int64_t C, int64_t K, int64_t HW) {
    CHECK_NQ(top1); CHECK_NQ(gt);
    TORCH_CHECK(top1.dtype() == torch::kInt64 && gt.dtype() == torch::kInt64,
                "enqueue_rows: int64 indices expected");
    const int B = gt.size(0);
    TORCH_CHECK(top1.size(0) == B && top1.size(1) == C * K,
                "top1 must be [B, C*K]");
    TORCH_CHECK(B <= 1024 && K <= 32, "envelope: B <= 1024, K <= 32");
    auto rows = torch::empty({B * K}, gt.options());
    auto lab = torch::empty({B * K}, gt.options());
    auto stream = at::hip::getCurrentHIPStream();
    const size_t lds = 2 * B * sizeof(int);
    hipLaunchKernelGGL(enqueue_rows_kernel, dim3(1), dim3(256), lds, stream,
                       top1.data_ptr<int64_t>(), gt.data_ptr<int64_t>(),
                       rows.data_ptr<int64_t>(), lab.data_ptr<int64_t>(),
                       B, (int)C, (int)K, (int)HW);
    return {rows, lab};
}

void bank_push(torch::Tensor feats, torch::Tensor labels, torch::Tensor mem,
               torch::Tensor head, torch::Tensor mem_len,
               int64_t C, int64_t cap) {
    CHECK_NQ(feats); CHECK_NQ(labels); CHECK_NQ(mem);
    CHECK_NQ(head); CHECK_NQ(mem_len);
    TORCH_CHECK(feats.dtype() == torch::kFloat32, "bank_push: fp32 feats");
    TORCH_CHECK(labels.dtype() == torch::kInt64, "bank_push: int64 labels");
    const int M = feats.size(0), d = feats.size(1);
    TORCH_CHECK(labels.size(0) == M, "labels must be [M]");
    TORCH_CHECK(mem.size(0) >= (C + 1) * cap && mem.size(1) == d,
                "mem must be [(C+1)*cap, d]");
    if (M == 0) return;
    auto stream = at::hip::getCurrentHIPStream();
    // one block per REAL class; sentinel-labelled rows are simply never
    // written (the torch path's trash row is unobservable anyway)
    hipLaunchKernelGGL(bank_push_kernel, dim3((int)C), dim3(256), 0,
                       stream, feats.data_ptr<float>(),
                       labels.data_ptr<int64_t>(), mem.data_ptr<float>(),
                       head.data_ptr<int64_t>(), mem_len.data_ptr<int64_t>(),
                       M, (int)cap, d);
}
