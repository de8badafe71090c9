// MI355X (gfx950 / CDNA4) scheduler kernels — the decision-plane hot loop.
//
// Semantics oracle: volcano_amd/ops/reference.py (PyTorch).  Reference
// behavior being rebuilt: volcano-sh/volcano pkg/scheduler/actions/allocate
// (allocate.go:719-953), util/predicate_helper.go:45, scheduler_helper.go:78,
// framework/statement.go (gang transaction) — re-designed as dense tensor
// passes instead of per-(task,node) Go callbacks.
//
// Layout: node state is SoA / transposed [R, N] so a wave's 64 lanes read
// 64 consecutive floats of one resource dim — fully coalesced on HBM3E.
// All kernels are enqueued on one HIP stream with ZERO host syncs per
// scheduling cycle; the gang commit/revert decision is taken on-device.
//
// Sizing: N (nodes) up to ~1M, R (resource dims) <= 16, W (label planes
// words) <= 4.  The per-class tensors are tiny (N*4B score) and L2/LLC
// resident; kernels are memory-latency/launch bound, not HBM bound.

#include <hip/hip_runtime.h>
#include <cfloat>
#include <cstdint>

#define DEVINL __device__ __forceinline__

namespace vamd {

constexpr int WAVE = 64;              // CDNA wavefront width (NOT 32)
#define NEG_INF (-__builtin_inff())   // matches the torch reference's -inf
constexpr float EPS = 1e-4f;
constexpr long long BIG_CAP = 1 << 30;

// ---------------------------------------------------------------------------
// K1+K2 fused: per-node feasibility + capacity + score for one task class.
//   grid-stride over N; one thread = one node; all loads coalesced ([R,N]).
// ---------------------------------------------------------------------------
__global__ void score_cap_kernel(
    const float* __restrict__ alloc,   // [R, N]
    const float* __restrict__ used,    // [R, N]
    const float* __restrict__ extra,   // [R, N] (releasing - pipelined) or nullptr
    const uint8_t* __restrict__ ready, // [N]
    const int64_t* __restrict__ taints,// [N]
    const int64_t* __restrict__ planes,// [W, N]
    const float* __restrict__ req,     // [R]
    int64_t tolerated,
    const int64_t* __restrict__ require, // [W]
    const int64_t* __restrict__ forbid,  // [W]
    float w_least, float w_most, float w_bal,
    const float* __restrict__ dim_w,   // [R]
    const float* __restrict__ bias,    // [N] or nullptr
    float* __restrict__ score_out,     // [N]
    int* __restrict__ cap_out,         // [N]
    int N, int R, int W)
{
    // req/dim_w are tiny and wave-uniform: scalar-cached by L1 automatically.
    float wsum = 0.f;
    for (int r = 0; r < R; ++r) wsum += dim_w[r];
    wsum = fmaxf(wsum, EPS);

    for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < N;
         i += gridDim.x * blockDim.x) {
        bool feasible = ready[i] != 0;
        feasible = feasible && ((taints[i] & ~tolerated) == 0);
        for (int w = 0; w < W; ++w) {
            int64_t p = planes[(size_t)w * N + i];
            feasible = feasible && ((p & require[w]) == require[w]);
            feasible = feasible && ((p & forbid[w]) == 0);
        }

        long long cap = BIG_CAP;
        float least = 0.f, most = 0.f, mean = 0.f;
        float frac[16];
        for (int r = 0; r < R; ++r) {
            size_t off = (size_t)r * N + i;
            float a = alloc[off];
            float u = used[off];
            float e = extra ? extra[off] : 0.f;
            float avail = a - u + e;
            float rq = req[r];
            if (rq > EPS) {
                feasible = feasible && (avail + EPS >= rq);
                float cc = (avail + EPS) / rq;     // clamp before the int
                if (cc > (float)BIG_CAP) cc = (float)BIG_CAP;  // cast: UB on inf
                long long c = (long long)floorf(cc);
                cap = min(cap, max(c, 0ll));
            }
            float f = fminf((u + rq) / fmaxf(a, EPS), 1.0f);
            frac[r] = f;
            float dw = dim_w[r];
            least += (1.0f - f) * dw;
            most += f * dw;
            mean += f;
        }
        mean /= (float)R;
        float var = 0.f;
        for (int r = 0; r < R; ++r) {
            float d = frac[r] - mean;
            var += d * d;
        }
        float bal = 1.0f - sqrtf(var / (float)R);
        float s = w_least * least / wsum + w_most * most / wsum + w_bal * bal;
        if (bias) s += bias[i];

        bool ok = feasible && cap > 0;
        score_out[i] = ok ? s : NEG_INF;
        cap_out[i] = ok ? (int)min(cap, BIG_CAP) : 0;
    }
}

// ---------------------------------------------------------------------------
// (value, index) argmax with ties -> lowest index, across a wave then a block.
// ---------------------------------------------------------------------------
struct ValIdx { float v; int i; };

DEVINL ValIdx better(ValIdx a, ValIdx b) {
    if (b.v > a.v || (b.v == a.v && b.i < a.i)) return b;
    return a;
}

DEVINL ValIdx wave_reduce(ValIdx x) {
    for (int off = WAVE / 2; off > 0; off >>= 1) {
        ValIdx o;
        o.v = __shfl_down(x.v, off);
        o.i = __shfl_down(x.i, off);
        x = better(x, o);
    }
    return x;
}

// ---------------------------------------------------------------------------
// K2 select + K4 gang-transactional placement for one task class.
//   ONE workgroup (1024 threads = 16 waves).  Iterative argmax over the
//   score vector with per-thread cached local maxima: after node b is
//   consumed only its owner thread rescans its stride (N/1024 elements,
//   L2-resident).  Updates used/queue_alloc/job accounting in place and
//   writes an undo log; optionally performs the single-class gang revert
//   in-kernel (fuse_min >= 0).
// ---------------------------------------------------------------------------
#define SC_THREADS 1024
#define SC_WAVES (SC_THREADS / WAVE)

__global__ void __launch_bounds__(SC_THREADS)
select_commit_kernel(
    float* __restrict__ score,        // [N] (consumed: selected entries -> -inf)
    const int* __restrict__ cap,      // [N]
    const float* __restrict__ req,    // [R]
    int ntasks,
    float* __restrict__ used,         // [R, N] in place
    float* __restrict__ queue_alloc,  // [R] in place
    const float* __restrict__ queue_limit, // [R]
    int* __restrict__ log_nodes,      // [K]
    int* __restrict__ log_counts,     // [K]
    int* __restrict__ log_len,        // [1]
    int* __restrict__ placed,         // [1]
    int* __restrict__ job_placed,     // [1]
    int fuse_min,                     // >=0: single-class gang min (in-kernel revert)
    int N, int R, int K)
{
    __shared__ ValIdx s_wave[SC_WAVES];
    __shared__ ValIdx s_best;
    __shared__ int s_remaining;
    __shared__ int s_cursor;
    __shared__ int s_take;
    __shared__ long long s_budget;

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;

    if (tid == 0) {
        // queue quota -> instance budget (proportion/capacity Allocatable
        // check, session_plugins.go:405 — enforced exactly, on device)
        long long quota = BIG_CAP;
        for (int r = 0; r < R; ++r) {
            float rq = req[r];
            if (rq > EPS) {
                float head = queue_limit[r] - queue_alloc[r];
                float qq = (head + EPS) / rq;
                if (qq > (float)BIG_CAP) qq = (float)BIG_CAP;
                long long q = (long long)floorf(qq);
                quota = min(quota, max(q, 0ll));
            }
        }
        s_budget = min((long long)ntasks, quota);
        s_remaining = (int)min(s_budget, (long long)INT32_MAX);
        s_cursor = 0;
    }
    __syncthreads();

    // per-thread local argmax over the strided slice {tid, tid+1024, ...}
    ValIdx loc; loc.v = NEG_INF; loc.i = INT32_MAX;
    for (int i = tid; i < N; i += SC_THREADS) {
        ValIdx c; c.v = score[i]; c.i = i;
        loc = better(loc, c);
    }

    while (true) {
        // block argmax: wave shuffle reduce, then wave 0 reduces the 16
        ValIdx w = wave_reduce(loc);
        if (lane == 0) s_wave[wid] = w;
        __syncthreads();
        if (wid == 0) {
            ValIdx b; b.v = NEG_INF; b.i = INT32_MAX;
            if (lane < SC_WAVES) b = s_wave[lane];
            b = wave_reduce(b);
            if (lane == 0) s_best = b;
        }
        __syncthreads();

        ValIdx best = s_best;
        if (best.v == NEG_INF || s_remaining <= 0 || s_cursor >= K) break;

        // take is published through LDS: every thread reading s_remaining
        // directly would race with thread 0's decrement below
        if (tid == 0) {
            s_take = min(cap[best.i], s_remaining);
            log_nodes[s_cursor] = best.i;
            log_counts[s_cursor] = s_take;
            s_cursor += 1;
            s_remaining -= s_take;
        }
        __syncthreads();
        int take = s_take;
        if (tid < R) {
            used[(size_t)tid * N + best.i] += (float)take * req[tid];
        }
        // knock out the chosen node; its owner rescans its slice
        if (tid == (best.i & (SC_THREADS - 1))) {
            score[best.i] = NEG_INF;
            loc.v = NEG_INF; loc.i = INT32_MAX;
            for (int i = tid; i < N; i += SC_THREADS) {
                ValIdx c; c.v = score[i]; c.i = i;
                loc = better(loc, c);
            }
        }
        __syncthreads();
    }

    __syncthreads();
    int total = (int)s_budget - s_remaining;
    bool revert = (fuse_min >= 0) && (total < fuse_min);

    if (revert) {
        // single-class gang failed: undo in place (statement.go:375 Discard)
        for (int e = wid; e < s_cursor; e += SC_WAVES) {
            int node = log_nodes[e];
            int cnt = log_counts[e];
            if (lane < R) used[(size_t)lane * N + node] -= (float)cnt * req[lane];
            if (lane == 0) log_counts[e] = 0;
        }
        __syncthreads();
        if (tid == 0) { *log_len = s_cursor; *placed = 0; }
    } else {
        if (tid < R) queue_alloc[tid] += (float)total * req[tid];
        if (tid == 0) {
            *log_len = s_cursor;
            *placed = total;
            *job_placed += total;
        }
    }
}

// ---------------------------------------------------------------------------
// K4 gang readiness for multi-class jobs (JobReady AND-aggregation,
// session_plugins.go:483).  One tiny block.
// ---------------------------------------------------------------------------
__global__ void finalize_job_kernel(
    const int* __restrict__ job_placed,   // [1]
    int occupied, int min_available,
    const int* __restrict__ class_placed, // [nc] (slice)
    const int* __restrict__ class_min,    // [nc]
    uint8_t* __restrict__ flag,           // [1]
    int nc)
{
    if (threadIdx.x == 0) {
        bool ok = (*job_placed + occupied) >= min_available;
        for (int c = 0; c < nc && ok; ++c)
            ok = class_placed[c] >= class_min[c];
        *flag = ok ? 1 : 0;
    }
}

// ---------------------------------------------------------------------------
// K4 conditional rollback of one class, gated on the job's device flag.
// ---------------------------------------------------------------------------
__global__ void cond_revert_kernel(
    const uint8_t* __restrict__ flag,  // [1] 1 = keep
    int* __restrict__ log_nodes,       // [K]
    int* __restrict__ log_counts,      // [K]
    const int* __restrict__ log_len,   // [1]
    const float* __restrict__ req,     // [R]
    float* __restrict__ used,          // [R, N]
    float* __restrict__ queue_alloc,   // [R]
    int* __restrict__ placed,          // [1]
    int* __restrict__ job_placed,      // [1]
    int N, int R)
{
    if (*flag) return;
    int len = *log_len;
    int tid = blockIdx.x * blockDim.x + threadIdx.x;
    int lane_r = tid % 16;           // up to 16 dims handled per entry
    int entry = tid / 16;
    int stride = (gridDim.x * blockDim.x) / 16;
    for (int e = entry; e < len; e += stride) {
        int cnt = log_counts[e];
        if (cnt == 0) continue;
        int node = log_nodes[e];
        if (lane_r < R)
            used[(size_t)lane_r * N + node] -= (float)cnt * req[lane_r];
    }
    if (tid == 0) {
        int p = *placed;
        for (int r = 0; r < R; ++r) queue_alloc[r] -= (float)p * req[r];
        *job_placed -= p;
        *placed = 0;
    }
    // zero the counts after the subtraction pass completes (same block
    // ordering is not guaranteed across blocks — run with ONE block)
    __syncthreads();
    for (int e = threadIdx.x; e < len; e += blockDim.x) log_counts[e] = 0;
}

}  // namespace vamd

// ---------------------------------------------------------------------------
// C ABI launchers (bound in bindings.cpp; stream passed from torch)
// ---------------------------------------------------------------------------
extern "C" {

void vamd_score_cap(
    const float* alloc, const float* used, const float* extra,
    const uint8_t* ready, const int64_t* taints, const int64_t* planes,
    const float* req, int64_t tolerated, const int64_t* require,
    const int64_t* forbid, float w_least, float w_most, float w_bal,
    const float* dim_w, const float* bias, float* score_out, int* cap_out,
    int N, int R, int W, hipStream_t stream)
{
    int threads = 256;
    int blocks = min((N + threads - 1) / threads, 4096);
    if (blocks < 1) blocks = 1;
    hipLaunchKernelGGL(vamd::score_cap_kernel, dim3(blocks), dim3(threads), 0,
                       stream, alloc, used, extra, ready, taints, planes, req,
                       tolerated, require, forbid, w_least, w_most, w_bal,
                       dim_w, bias, score_out, cap_out, N, R, W);
}

void vamd_select_commit(
    float* score, const int* cap, const float* req, int ntasks, float* used,
    float* queue_alloc, const float* queue_limit, int* log_nodes,
    int* log_counts, int* log_len, int* placed, int* job_placed, int fuse_min,
    int N, int R, int K, hipStream_t stream)
{
    hipLaunchKernelGGL(vamd::select_commit_kernel, dim3(1), dim3(SC_THREADS),
                       0, stream, score, cap, req, ntasks, used, queue_alloc,
                       queue_limit, log_nodes, log_counts, log_len, placed,
                       job_placed, fuse_min, N, R, K);
}

void vamd_finalize_job(
    const int* job_placed, int occupied, int min_available,
    const int* class_placed, const int* class_min, uint8_t* flag, int nc,
    hipStream_t stream)
{
    hipLaunchKernelGGL(vamd::finalize_job_kernel, dim3(1), dim3(64), 0,
                       stream, job_placed, occupied, min_available,
                       class_placed, class_min, flag, nc);
}

void vamd_cond_revert(
    const uint8_t* flag, int* log_nodes, int* log_counts, const int* log_len,
    const float* req, float* used, float* queue_alloc, int* placed,
    int* job_placed, int N, int R, hipStream_t stream)
{
    hipLaunchKernelGGL(vamd::cond_revert_kernel, dim3(1), dim3(1024), 0,
                       stream, flag, log_nodes, log_counts, log_len, req,
                       used, queue_alloc, placed, job_placed, N, R);
}

}  // extern "C"
