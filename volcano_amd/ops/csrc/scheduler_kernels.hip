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

#include "vamd_api.h"   // VamdClassDesc/VamdJobDesc for the megacycle

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
// core of K1+K2: score/cap for nodes {start, start+step, ...} — shared
// by the grid-stride kernel and the fused megacycle (block stride)
// per-node score/cap/feasibility — the ONE implementation shared by the
// grid-stride kernel, the batched chain scorer, and the chain select's
// live re-scoring of touched nodes.  Float op order is fixed: every
// caller gets bit-identical values for identical inputs (the decision-
// exactness contract with the torch oracle).
struct NodeScore { float s; int cap; };

DEVINL NodeScore node_score_one(
    int i,
    const float* __restrict__ alloc,
    const float* __restrict__ used,
    const float* __restrict__ extra,
    const uint8_t* __restrict__ ready,
    const int64_t* __restrict__ taints,
    const int64_t* __restrict__ planes,
    const float* __restrict__ req,
    int64_t tolerated,
    const int64_t* __restrict__ require,
    const int64_t* __restrict__ forbid,
    float w_least, float w_most, float w_bal,
    const float* __restrict__ dim_w, float wsum,
    const float* __restrict__ bias,
    int N, int R, int W)
{
    bool feasible = ready[i] != 0;
    feasible = feasible && ((taints[i] & ~tolerated) == 0);
    for (int w = 0; w < W; ++w) {
        int64_t p = planes[(size_t)w * N + i];
        feasible = feasible && ((p & require[w]) == require[w]);
        feasible = feasible && ((p & forbid[w]) == 0);
    }

    long long cap = BIG_CAP;
    float least = 0.f, most = 0.f, mean = 0.f;
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
        float dw = dim_w[r];
        least += (1.0f - f) * dw;
        most += f * dw;
        mean += f;
    }
    mean /= (float)R;
    // balanced-allocation variance: second pass RECOMPUTES f instead
    // of staging a per-thread frac[] array (a fixed-size array caps R
    // and spills VGPRs; inputs are L2-resident so the re-read is
    // cheap, and the float op sequence is identical — bit-equal to
    // the torch oracle).  Skipped entirely when the weight is 0.
    float bal = 0.f;
    if (w_bal != 0.f) {
        float var = 0.f;
        for (int r = 0; r < R; ++r) {
            size_t off = (size_t)r * N + i;
            float f = fminf((used[off] + req[r]) /
                            fmaxf(alloc[off], EPS), 1.0f);
            float d = f - mean;
            var += d * d;
        }
        bal = 1.0f - sqrtf(var / (float)R);
    }
    float s = w_least * least / wsum + w_most * most / wsum + w_bal * bal;
    if (bias) s += bias[i];

    bool ok = feasible && cap > 0;
    NodeScore out;
    out.s = ok ? s : NEG_INF;
    out.cap = ok ? (int)min(cap, BIG_CAP) : 0;
    return out;
}

__device__ void score_pass(
    int start, int step,
    const float* __restrict__ alloc,
    const float* __restrict__ used,
    const float* __restrict__ extra,
    const uint8_t* __restrict__ ready,
    const int64_t* __restrict__ taints,
    const int64_t* __restrict__ planes,
    const float* __restrict__ req,
    int64_t tolerated,
    const int64_t* __restrict__ require,
    const int64_t* __restrict__ forbid,
    float w_least, float w_most, float w_bal,
    const float* __restrict__ dim_w,
    const float* __restrict__ bias,
    float* __restrict__ score_out,
    int* __restrict__ cap_out,
    int N, int R, int W)
{
    // req/dim_w are tiny and wave-uniform: scalar-cached by L1 automatically.
    float wsum = 0.f;
    for (int r = 0; r < R; ++r) wsum += dim_w[r];
    wsum = fmaxf(wsum, EPS);

    for (int i = start; i < N; i += step) {
        NodeScore ns = node_score_one(
            i, alloc, used, extra, ready, taints, planes, req, tolerated,
            require, forbid, w_least, w_most, w_bal, dim_w, wsum, bias,
            N, R, W);
        score_out[i] = ns.s;
        if (cap_out) cap_out[i] = ns.cap;
    }
}

__global__ void score_cap_kernel(
    const float* alloc, const float* used, const float* extra,
    const uint8_t* ready, const int64_t* taints, const int64_t* planes,
    const float* req, int64_t tolerated, const int64_t* require,
    const int64_t* forbid, float w_least, float w_most, float w_bal,
    const float* dim_w, const float* bias, float* score_out, int* cap_out,
    int N, int R, int W)
{
    score_pass(blockIdx.x * blockDim.x + threadIdx.x,
               gridDim.x * blockDim.x,
               alloc, used, extra, ready, taints, planes, req, tolerated,
               require, forbid, w_least, w_most, w_bal, dim_w, bias,
               score_out, cap_out, N, R, W);
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

__device__ void dev_select_commit(
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

__global__ void __launch_bounds__(SC_THREADS)
select_commit_kernel(
    float* score, const int* cap, const float* req, int ntasks, float* used,
    float* queue_alloc, const float* queue_limit, int* log_nodes,
    int* log_counts, int* log_len, int* placed, int* job_placed,
    int fuse_min, int N, int R, int K)
{
    dev_select_commit(score, cap, req, ntasks, used, queue_alloc,
                      queue_limit, log_nodes, log_counts, log_len, placed,
                      job_placed, fuse_min, N, R, K);
}

// ---------------------------------------------------------------------------
// Fused score+select for small classes: ONE launch does the score pass
// (block-stride, 16 waves hide the L2 latency at small N) and the
// serial select.  Halves the per-class launch count on heterogeneous
// plans where the grid-stride score kernel was dispatch-latency bound
// (~12.5 us/class -> ~7 us measured on the mix bench).  Bit-identical
// math: same score_pass + dev_select_commit bodies.
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(SC_THREADS)
fused_score_select_kernel(
    const float* __restrict__ alloc, float* __restrict__ used,
    const float* __restrict__ extra,
    const uint8_t* __restrict__ ready, const int64_t* __restrict__ taints,
    const int64_t* __restrict__ planes,
    const float* __restrict__ req, int64_t tolerated,
    const int64_t* __restrict__ require, const int64_t* __restrict__ forbid,
    float w_least, float w_most, float w_bal,
    const float* __restrict__ dim_w, const float* __restrict__ bias,
    float* __restrict__ score, int* __restrict__ cap,
    int ntasks,
    float* __restrict__ queue_alloc, const float* __restrict__ queue_limit,
    int* __restrict__ log_nodes, int* __restrict__ log_counts,
    int* __restrict__ log_len, int* __restrict__ placed,
    int* __restrict__ job_placed, int fuse_min,
    int N, int R, int W, int K)
{
    score_pass(threadIdx.x, SC_THREADS, alloc, used, extra, ready, taints,
               planes, req, tolerated, require, forbid,
               w_least, w_most, w_bal, dim_w, bias, score, cap, N, R, W);
    __syncthreads();
    dev_select_commit(score, cap, req, ntasks, used, queue_alloc,
                      queue_limit, log_nodes, log_counts, log_len, placed,
                      job_placed, fuse_min, N, R, K);
}

// ---------------------------------------------------------------------------
// K4 gang readiness for multi-class jobs (JobReady AND-aggregation,
// session_plugins.go:483).  One tiny block.
// ---------------------------------------------------------------------------
DEVINL void dev_finalize_job(
    const int* job_placed, int occupied, int min_available,
    const int* class_placed, const int* class_min, uint8_t* flag, int nc)
{
    if (threadIdx.x == 0) {
        bool ok = (*job_placed + occupied) >= min_available;
        for (int c = 0; c < nc && ok; ++c)
            ok = class_placed[c] >= class_min[c];
        *flag = ok ? 1 : 0;
    }
}

__global__ void finalize_job_kernel(
    const int* __restrict__ job_placed,   // [1]
    int occupied, int min_available,
    const int* __restrict__ class_placed, // [nc] (slice)
    const int* __restrict__ class_min,    // [nc]
    uint8_t* __restrict__ flag,           // [1]
    int nc)
{
    dev_finalize_job(job_placed, occupied, min_available, class_placed,
                     class_min, flag, nc);
}

// ---------------------------------------------------------------------------
// K4 conditional rollback of one class, gated on the job's device flag.
// Block-scope (callers launch ONE block / run inside the megacycle).
// ---------------------------------------------------------------------------
__device__ void dev_cond_revert(
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
    int tid = threadIdx.x;
    int lane_r = tid % WAVE;         // one wave per entry: R <= 64 dims
    int entry = tid / WAVE;
    int stride = blockDim.x / WAVE;
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
    // zero the counts after the subtraction pass completes
    __syncthreads();
    for (int e = threadIdx.x; e < len; e += blockDim.x) log_counts[e] = 0;
}

__global__ void cond_revert_kernel(
    const uint8_t* flag, int* log_nodes, int* log_counts,
    const int* log_len, const float* req, float* used, float* queue_alloc,
    int* placed, int* job_placed, int N, int R)
{
    dev_cond_revert(flag, log_nodes, log_counts, log_len, req, used,
                    queue_alloc, placed, job_placed, N, R);
}

// ---------------------------------------------------------------------------
// K2/K4 bulk select for mega-classes (gang bundles): the serial argmax
// loop above costs one block-reduce + owner rescan PER CONSUMED NODE
// (~10k sequential steps for a 100k-task bundle).  Scores are static
// within a class, so the fill equals "stable-sort nodes by score desc,
// take caps until the budget" — exactly the torch oracle
// (reference.py select_commit: argsort + cumsum + clamp).  This kernel
// computes that directly: a workgroup LSD radix sort (8 passes of 4-bit
// digits, rank-stable) followed by a chunked block scan that emits the
// undo log in sorted order.  One workgroup; ~16N global accesses total
// instead of ~N per consumed node.
// ---------------------------------------------------------------------------
#define BS_THREADS 1024
#define BS_WAVES (BS_THREADS / WAVE)   // 16
#define BS_DIGITS 16                   // 4-bit LSD
#define BS_PASSES 8

// monotonic float->uint map, then invert so ASCENDING key == score DESC
DEVINL unsigned score_key(float s) {
    unsigned u = __float_as_uint(s);
    unsigned m = (u >> 31) ? 0xFFFFFFFFu : 0x80000000u;
    return ~(u ^ m);
}

__global__ void __launch_bounds__(BS_THREADS)
bulk_select_kernel(
    const float* __restrict__ score,       // [N]
    const int* __restrict__ cap,           // [N]
    const float* __restrict__ req,         // [R]
    int ntasks,
    float* __restrict__ used,              // [R, N] in place
    float* __restrict__ queue_alloc,       // [R] in place
    const float* __restrict__ queue_limit, // [R]
    int* __restrict__ log_nodes,           // [K]
    int* __restrict__ log_counts,          // [K]
    int* __restrict__ log_len,             // [1]
    int* __restrict__ placed,              // [1]
    int* __restrict__ job_placed,          // [1]
    int fuse_min,
    unsigned* __restrict__ scratch,        // [4N]: keys/ids ping-pong
    int N, int R, int K)
{
    __shared__ unsigned s_hist[BS_DIGITS * BS_THREADS];     // 64 KB
    __shared__ unsigned s_digit_base[BS_DIGITS];
    __shared__ long long s_budget;
    __shared__ long long s_carry_cap;      // running csum across chunks
    __shared__ int s_carry_log;            // running log-entry count
    __shared__ long long s_wave_ll[BS_WAVES];
    __shared__ int s_wave_i[BS_WAVES];

    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;

    unsigned* keys_a = scratch;
    unsigned* ids_a = scratch + (size_t)N;
    unsigned* keys_b = scratch + 2 * (size_t)N;
    unsigned* ids_b = scratch + 3 * (size_t)N;

    if (tid == 0) {
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
        s_carry_cap = 0;
        s_carry_log = 0;
    }

    // -- build (key, id) pairs; array position i = node i (stable base) --
    for (int i = tid; i < N; i += BS_THREADS) {
        keys_a[i] = score_key(score[i]);
        ids_a[i] = (unsigned)i;
    }
    __syncthreads();

    // -- 8-pass LSD radix (each thread owns a CONTIGUOUS chunk: rank
    //    order inside a digit bin is (thread, position) = input order,
    //    which is what makes the sort stable) -------------------------------
    const int chunk = (N + BS_THREADS - 1) / BS_THREADS;
    const int lo = min(tid * chunk, N);
    const int hi = min(lo + chunk, N);

    for (int pass = 0; pass < BS_PASSES; ++pass) {
        unsigned* in_k = (pass & 1) ? keys_b : keys_a;
        unsigned* in_i = (pass & 1) ? ids_b : ids_a;
        unsigned* out_k = (pass & 1) ? keys_a : keys_b;
        unsigned* out_i = (pass & 1) ? ids_a : ids_b;
        const int shift = pass * 4;

        unsigned cnt[BS_DIGITS];
        for (int d = 0; d < BS_DIGITS; ++d) cnt[d] = 0;
        for (int i = lo; i < hi; ++i)
            cnt[(in_k[i] >> shift) & 15u]++;
        for (int d = 0; d < BS_DIGITS; ++d)
            s_hist[d * BS_THREADS + tid] = cnt[d];
        __syncthreads();

        // wave `wid` scans digit row `wid` (inclusive, 16 chunks of 64)
        {
            unsigned carry = 0;
            unsigned* row = s_hist + wid * BS_THREADS;
            for (int c = 0; c < BS_THREADS / WAVE; ++c) {
                unsigned v = row[c * WAVE + lane];
                for (int off = 1; off < WAVE; off <<= 1) {
                    unsigned o = __shfl_up(v, off);
                    if (lane >= off) v += o;
                }
                v += carry;
                row[c * WAVE + lane] = v;
                carry = __shfl(v, WAVE - 1);
            }
            if (lane == 0) s_digit_base[wid] = carry;   // digit total
        }
        __syncthreads();
        if (tid == 0) {
            unsigned run = 0;
            for (int d = 0; d < BS_DIGITS; ++d) {
                unsigned t = s_digit_base[d];
                s_digit_base[d] = run;
                run += t;
            }
        }
        __syncthreads();

        unsigned off[BS_DIGITS];
        for (int d = 0; d < BS_DIGITS; ++d)
            off[d] = s_digit_base[d] + s_hist[d * BS_THREADS + tid] - cnt[d];
        for (int i = lo; i < hi; ++i) {
            unsigned k = in_k[i];
            unsigned d = (k >> shift) & 15u;
            unsigned p = off[d]++;
            out_k[p] = k;
            out_i[p] = in_i[i];
        }
        __syncthreads();
    }
    // even pass count -> sorted result is back in keys_a / ids_a

    // -- chunked block scan over the sorted order: cumulative caps ->
    //    per-position take, log compaction, usage commit ---------------------
    const long long budget = s_budget;
    const int nchunks = (N + BS_THREADS - 1) / BS_THREADS;
    long long total_ll = 0;

    for (int c = 0; c < nchunks; ++c) {
        int j = c * BS_THREADS + tid;
        long long cj = 0;
        int node = -1;
        if (j < N) {
            node = (int)ids_a[j];
            float s = score[node];
            if (s != NEG_INF) cj = (long long)cap[node];
        }
        // block inclusive scan of cj (wave scan + cross-wave offsets)
        long long v = cj;
        for (int off = 1; off < WAVE; off <<= 1) {
            long long o = __shfl_up(v, off);
            if (lane >= off) v += o;
        }
        if (lane == WAVE - 1) s_wave_ll[wid] = v;
        __syncthreads();
        if (wid == 0) {
            long long w = (lane < BS_WAVES) ? s_wave_ll[lane] : 0;
            for (int off = 1; off < BS_WAVES; off <<= 1) {
                long long o = __shfl_up(w, off);
                if (lane >= off) w += o;
            }
            if (lane < BS_WAVES) s_wave_ll[lane] = w;
        }
        __syncthreads();
        long long csum = v + (wid > 0 ? s_wave_ll[wid - 1] : 0) + s_carry_cap;

        long long over = csum - budget; if (over < 0) over = 0;
        long long take_ll = cj - over; if (take_ll < 0) take_ll = 0;
        int take = (int)take_ll;
        int flag = take > 0 ? 1 : 0;

        // block inclusive scan of flags -> log slot
        int f = flag;
        for (int off = 1; off < WAVE; off <<= 1) {
            int o = __shfl_up(f, off);
            if (lane >= off) f += o;
        }
        if (lane == WAVE - 1) s_wave_i[wid] = f;
        __syncthreads();
        if (wid == 0) {
            int w = (lane < BS_WAVES) ? s_wave_i[lane] : 0;
            for (int off = 1; off < BS_WAVES; off <<= 1) {
                int o = __shfl_up(w, off);
                if (lane >= off) w += o;
            }
            if (lane < BS_WAVES) s_wave_i[lane] = w;
        }
        __syncthreads();
        int fsum = f + (wid > 0 ? s_wave_i[wid - 1] : 0) + s_carry_log;

        if (flag && fsum - 1 < K) {
            log_nodes[fsum - 1] = node;
            log_counts[fsum - 1] = take;
        }
        __syncthreads();
        if (tid == BS_THREADS - 1) {
            s_carry_cap = csum;
            s_carry_log = fsum;
        }
        __syncthreads();
    }
    total_ll = min(s_carry_cap, budget);
    int total = (int)total_ll;
    int m = min(s_carry_log, K);
    bool revert = (fuse_min >= 0) && (total < fuse_min);

    if (revert) {
        // gang can't reach its minimum: nothing is applied; the log keeps
        // zeroed counts (same end state as the serial kernel's discard)
        for (int e = tid; e < m; e += BS_THREADS) log_counts[e] = 0;
        if (tid == 0) { *log_len = m; *placed = 0; }
        return;
    }
    // commit: apply the logged takes to the staged usage
    for (int e = wid; e < m; e += BS_WAVES) {
        int node = log_nodes[e];
        int cnt = log_counts[e];
        if (lane < R) used[(size_t)lane * N + node] += (float)cnt * req[lane];
    }
    if (tid < R) queue_alloc[tid] += (float)total * req[tid];
    if (tid == 0) {
        *log_len = m;
        *placed = total;
        *job_placed += total;
    }
}

// ---------------------------------------------------------------------------
// Megacycle: a WHOLE small-class plan in ONE launch.  Heterogeneous
// inventories produce thousands of per-job classes; at ~2 enqueues per
// class the cycle becomes launch-bound.  This kernel sequences the
// per-job kernel bodies device-side (score -> select -> gang
// finalize/revert), preserving the exact sequential semantics: one
// workgroup, every class sees the previous classes' staged usage.
// Dispatched only when no class takes the bulk path (cycle_runner).
// ---------------------------------------------------------------------------
__global__ void __launch_bounds__(SC_THREADS)
megacycle_kernel(
    const VamdClassDesc* __restrict__ classes,   // [C] (device copy)
    const VamdJobDesc* __restrict__ jobs,        // [J] (device copy)
    int n_jobs,
    const float* __restrict__ alloc, float* __restrict__ used,
    const float* __restrict__ extra,
    const uint8_t* __restrict__ ready, const int64_t* __restrict__ taints,
    const int64_t* __restrict__ planes,
    const float* __restrict__ bias, const float* __restrict__ bias_rows,
    const float* __restrict__ class_req,        // [C, R]
    const int64_t* __restrict__ class_tol,      // [C] (device copy)
    const int64_t* __restrict__ class_require,  // [C, W]
    const int64_t* __restrict__ class_forbid,   // [C, W]
    const int32_t* __restrict__ class_min,      // [C]
    const float* __restrict__ dim_w,
    float* __restrict__ queue_alloc, const float* __restrict__ queue_limit,
    float* __restrict__ score, int* __restrict__ cap,
    int* __restrict__ log_nodes, int* __restrict__ log_counts,
    int* __restrict__ log_len, int* __restrict__ class_placed,
    int* __restrict__ job_placed, uint8_t* __restrict__ job_flag,
    int N, int R, int W)
{
    const int tid = threadIdx.x;
    for (int j = 0; j < n_jobs; ++j) {
        const VamdJobDesc jd = jobs[j];
        const int nc = jd.class_end - jd.class_begin;
        const bool single = (nc == 1);

        for (int c = jd.class_begin; c < jd.class_end; ++c) {
            const VamdClassDesc cd = classes[c];
            const float* ext = (cd.flags & 1) ? extra : nullptr;
            const float* b = (cd.bias_row >= 0 && bias_rows)
                ? bias_rows + (size_t)cd.bias_row * N : bias;

            score_pass(tid, SC_THREADS, alloc, used, ext, ready, taints,
                       planes, class_req + (size_t)c * R, class_tol[c],
                       class_require + (size_t)c * W,
                       class_forbid + (size_t)c * W,
                       cd.w_least, cd.w_most, cd.w_bal, dim_w, b,
                       score, cap, N, R, W);
            __syncthreads();

            int fuse_min = -1;
            if (single) {
                int need = jd.min_available - jd.occupied;
                if (cd.min_needed > need) need = cd.min_needed;
                fuse_min = need > 0 ? need : 0;
            }
            dev_select_commit(score, cap, class_req + (size_t)c * R,
                              cd.ntasks, used,
                              queue_alloc + (size_t)cd.queue_idx * R,
                              queue_limit + (size_t)cd.queue_idx * R,
                              log_nodes + cd.log_off,
                              log_counts + cd.log_off,
                              log_len + c, class_placed + c, job_placed + j,
                              fuse_min, N, R, cd.log_cap);
            __syncthreads();
        }

        if (!single) {
            dev_finalize_job(job_placed + j, jd.occupied, jd.min_available,
                             class_placed + jd.class_begin,
                             class_min + jd.class_begin, job_flag + j, nc);
            __syncthreads();
            for (int c = jd.class_begin; c < jd.class_end; ++c) {
                const VamdClassDesc cd = classes[c];
                dev_cond_revert(job_flag + j, log_nodes + cd.log_off,
                                log_counts + cd.log_off, log_len + c,
                                class_req + (size_t)c * R, used,
                                queue_alloc + (size_t)cd.queue_idx * R,
                                class_placed + c, job_placed + j, N, R);
                __syncthreads();
            }
        }
    }
}

// ---------------------------------------------------------------------------
// Chain path for many-small-class plans (heterogeneous mixes).  The
// per-class pipeline pays 2 launches + a serial under-occupied score
// pass per class (~10k classes in a mixed inventory).  Here a CHUNK of
// consecutive single-class jobs runs as TWO launches:
//
//   1. batch_score_kernel — scores ALL classes of the chunk against the
//      usage at chunk start, one class per blockIdx.y, filling the chip
//      (the [C,R]x[R,N] pass; L2-resident node planes make it bandwidth-
//      trivial).
//   2. select_chain_kernel — one workgroup replays the selects in job
//      order with LAZY EXACT re-scoring: a node whose usage changed
//      since the chunk snapshot ("touched") is re-scored live via the
//      same node_score_one; untouched nodes' snapshot scores ARE their
//      live scores.  The merged argmax is therefore bit-identical to
//      re-scoring every class against live usage — the chain is a pure
//      optimization, decisions match the per-class path and the torch
//      oracle exactly.
//
// Reference semantics covered: allocate.go:719-866 per-job sequential
// order; statement.go gang revert fused per single-class job.
// ---------------------------------------------------------------------------
__global__ void batch_score_kernel(
    const VamdClassDesc* __restrict__ classes,   // [C] device
    int c0,
    const float* __restrict__ alloc, const float* __restrict__ used,
    const float* __restrict__ extra,
    const uint8_t* __restrict__ ready, const int64_t* __restrict__ taints,
    const int64_t* __restrict__ planes,
    const float* __restrict__ bias, const float* __restrict__ bias_rows,
    const float* __restrict__ class_req, const int64_t* __restrict__ class_tol,
    const int64_t* __restrict__ class_require,
    const int64_t* __restrict__ class_forbid,
    const float* __restrict__ dim_w,
    float* __restrict__ score_buf,               // [chunk, N]
    int N, int R, int W)
{
    const int c = c0 + blockIdx.y;
    const VamdClassDesc cd = classes[c];
    const float* ext = (cd.flags & 1) ? extra : nullptr;
    const float* b = (cd.bias_row >= 0 && bias_rows)
        ? bias_rows + (size_t)cd.bias_row * N : bias;
    score_pass(blockIdx.x * blockDim.x + threadIdx.x,
               gridDim.x * blockDim.x,
               alloc, used, ext, ready, taints, planes,
               class_req + (size_t)c * R, class_tol[c],
               class_require + (size_t)c * W,
               class_forbid + (size_t)c * W,
               cd.w_least, cd.w_most, cd.w_bal, dim_w, b,
               score_buf + (size_t)blockIdx.y * N, nullptr, N, R, W);
}

#define CH_THREADS 256
#define CH_WAVES (CH_THREADS / WAVE)
#define CH_TOPK 16

// Per-class top-K candidates by snapshot score (desc, ties → lower
// index), computed in PARALLEL across the chunk (one block per class)
// so the serial select chain never scans the full node row in the
// common case: the best untouched node is the first untouched entry of
// the candidate list.  Chosen entries are knocked out of the row
// (-inf) — the chain handles them via the candidate list / touched set,
// and the rare fallback scan only needs the remaining nodes.
__global__ void __launch_bounds__(SC_THREADS)
topk_kernel(
    float* __restrict__ score_buf,        // [chunk, N] (chosen -> -inf)
    float* __restrict__ topk_vals,        // [chunk, CH_TOPK]
    int* __restrict__ topk_ids,           // [chunk, CH_TOPK]
    int N)
{
    __shared__ ValIdx s_wave[SC_WAVES];
    __shared__ ValIdx s_best;
    float* row = score_buf + (size_t)blockIdx.x * N;
    float* tv = topk_vals + (size_t)blockIdx.x * CH_TOPK;
    int* ti = topk_ids + (size_t)blockIdx.x * CH_TOPK;
    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;

    ValIdx loc; loc.v = NEG_INF; loc.i = INT32_MAX;
    for (int i = tid; i < N; i += SC_THREADS) {
        ValIdx c; c.v = row[i]; c.i = i;
        loc = better(loc, c);
    }
    for (int k = 0; k < CH_TOPK; ++k) {
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
        if (tid == 0) { tv[k] = best.v; ti[k] = best.i; }
        if (best.v == NEG_INF) {
            // fill the tail so the chain's walk can stop at -inf
            if (tid == 0)
                for (int t = k; t < CH_TOPK; ++t) { tv[t] = NEG_INF; ti[t] = INT32_MAX; }
            return;
        }
        if (tid == (best.i & (SC_THREADS - 1))) {
            row[best.i] = NEG_INF;
            loc.v = NEG_INF; loc.i = INT32_MAX;
            for (int i = tid; i < N; i += SC_THREADS) {
                ValIdx c; c.v = row[i]; c.i = i;
                loc = better(loc, c);
            }
        }
        __syncthreads();
    }
}

// 256 threads / 4 waves: enough waves to hide the L2 latency of the
// touched-list re-scoring (one wave measured 5x slower — latency
// bound), small enough that the two per-pop barriers stay cheap.
// Scalar decisions (budget/cap/take) are computed REDUNDANTLY by every
// thread from the same memory — uniform by construction, no broadcast
// barrier needed; the cross-wave argmax merge is one LDS write + one
// barrier + a 4-entry sequential merge on every thread.
__global__ void __launch_bounds__(CH_THREADS)
select_chain_kernel(
    const VamdClassDesc* __restrict__ classes,   // [C] device
    const VamdJobDesc* __restrict__ jobs,        // [J] device
    int c0, int c1,
    const float* __restrict__ alloc, float* __restrict__ used,
    const float* __restrict__ extra,
    const uint8_t* __restrict__ ready, const int64_t* __restrict__ taints,
    const int64_t* __restrict__ planes,
    const float* __restrict__ bias, const float* __restrict__ bias_rows,
    const float* __restrict__ class_req, const int64_t* __restrict__ class_tol,
    const int64_t* __restrict__ class_require,
    const int64_t* __restrict__ class_forbid,
    const float* __restrict__ dim_w,
    float* __restrict__ queue_alloc, const float* __restrict__ queue_limit,
    float* __restrict__ score_buf,               // [c1-c0, N] snapshot scores
    const float* __restrict__ topk_vals,         // [c1-c0, CH_TOPK]
    const int* __restrict__ topk_ids,            // [c1-c0, CH_TOPK]
    int* __restrict__ log_nodes, int* __restrict__ log_counts,
    int* __restrict__ log_len, int* __restrict__ class_placed,
    int* __restrict__ job_placed,
    uint8_t* __restrict__ touched,               // [N] scratch (reset here)
    int* __restrict__ touched_list,              // [N] scratch
    int N, int R, int W)
{
    __shared__ ValIdx s_wave[CH_WAVES];
    const int tid = threadIdx.x;
    const int lane = tid & (WAVE - 1);
    const int wid = tid / WAVE;

    // touched set is relative to THIS chunk's score snapshot
    for (int i = tid; i < N; i += CH_THREADS) touched[i] = 0;
    int tcount = 0;                 // block-uniform (all threads mirror it)
    __syncthreads();

    float wsum = 0.f;
    for (int r = 0; r < R; ++r) wsum += dim_w[r];
    wsum = fmaxf(wsum, EPS);

    for (int c = c0; c < c1; ++c) {
        const VamdClassDesc cd = classes[c];
        const VamdJobDesc jd = jobs[cd.job_idx];
        const float* req = class_req + (size_t)c * R;
        const float* ext = (cd.flags & 1) ? extra : nullptr;
        const float* b = (cd.bias_row >= 0 && bias_rows)
            ? bias_rows + (size_t)cd.bias_row * N : bias;
        const float* row = score_buf + (size_t)(c - c0) * N;
        float* qa = queue_alloc + (size_t)cd.queue_idx * R;
        const float* ql = queue_limit + (size_t)cd.queue_idx * R;
        int* ln = log_nodes + cd.log_off;
        int* lc = log_counts + cd.log_off;
        const int K = cd.log_cap;
        int fuse = jd.min_available - jd.occupied;
        if (cd.min_needed > fuse) fuse = cd.min_needed;
        if (fuse < 0) fuse = 0;

        // queue quota — per-dim candidates on lanes < R, wave min-reduce
        // (sequential dependent L2 loads cost ~300 ns each; the wave does
        // them in one round).  Every wave computes the same values from
        // the same memory → block-uniform without a barrier.
        long long qcand = BIG_CAP;
        if (lane < R) {
            float rq = req[lane];
            if (rq > EPS) {
                float head = ql[lane] - qa[lane];
                float qq = (head + EPS) / rq;
                if (qq > (float)BIG_CAP) qq = (float)BIG_CAP;
                long long q = (long long)floorf(qq);
                qcand = max(q, 0ll);
            }
        }
        for (int off = WAVE / 2; off > 0; off >>= 1)
            qcand = min(qcand, (long long)__shfl_down((long long)qcand, off));
        const long long quota = __shfl((long long)qcand, 0);
        const long long budget = min((long long)cd.ntasks, quota);
        int remaining = (int)min(budget, (long long)INT32_MAX);
        int cursor = 0;
        const float* tv = topk_vals + (size_t)(c - c0) * CH_TOPK;
        const int* ti = topk_ids + (size_t)(c - c0) * CH_TOPK;

        while (remaining > 0 && cursor < K) {
            // stale side: first UNTOUCHED candidate of the top-K list is
            // exactly the best untouched node (every node outside the
            // list scores <= the last candidate).  Lanes load the list in
            // parallel; a ballot finds the first untouched entry — the
            // serial walk paid ~16 dependent L2 loads here.
            float kv = NEG_INF;
            int ki = INT32_MAX;
            bool feas = false, ok = false;
            if (lane < CH_TOPK) {
                kv = tv[lane];
                ki = ti[lane];
                feas = (kv != NEG_INF);
                ok = feas && !touched[ki];
            }
            uint64_t m_ok = __ballot(ok);
            uint64_t m_feas = __ballot(feas);
            ValIdx stale; stale.v = NEG_INF; stale.i = INT32_MAX;
            bool fallback = false;
            if (m_ok != 0) {
                int pos = __ffsll((unsigned long long)m_ok) - 1;
                stale.v = __shfl(kv, pos);
                stale.i = __shfl(ki, pos);
            } else if ((m_feas & ((1ull << CH_TOPK) - 1))
                       == ((1ull << CH_TOPK) - 1)) {
                // all CH_TOPK candidates feasible AND touched: the best
                // untouched (if any) is below the list — full row scan
                fallback = true;
            }
            ValIdx loc; loc.v = NEG_INF; loc.i = INT32_MAX;
            if (fallback) {
                for (int i = tid; i < N; i += CH_THREADS) {
                    if (!touched[i]) {
                        ValIdx cnd; cnd.v = row[i]; cnd.i = i;
                        loc = better(loc, cnd);
                    }
                }
            }
            loc = better(loc, stale);   // uniform: merges identically
            // touched side: live re-score (usage changed since snapshot)
            for (int k = tid; k < tcount; k += CH_THREADS) {
                int i = touched_list[k];
                bool consumed = false;       // already taken by THIS class
                for (int e = 0; e < cursor; ++e)
                    if (ln[e] == i) { consumed = true; break; }
                if (consumed) continue;
                NodeScore ns = node_score_one(
                    i, alloc, used, ext, ready, taints, planes, req,
                    class_tol[c], class_require + (size_t)c * W,
                    class_forbid + (size_t)c * W,
                    cd.w_least, cd.w_most, cd.w_bal, dim_w, wsum, b,
                    N, R, W);
                ValIdx cnd; cnd.v = ns.s; cnd.i = i;
                loc = better(loc, cnd);
            }
            ValIdx w = wave_reduce(loc);
            if (lane == 0) s_wave[wid] = w;
            __syncthreads();                      // barrier 1
            ValIdx best = s_wave[0];              // 4-entry merge, uniform
            for (int q = 1; q < CH_WAVES; ++q) best = better(best, s_wave[q]);
            if (best.v == NEG_INF) break;

            // live capacity of the chosen node: per-dim candidates on
            // lanes < R, wave min-reduce (uniform across waves; untouched
            // nodes keep their snapshot capacity by definition)
            long long ccand = BIG_CAP;
            if (lane < R) {
                float rq = req[lane];
                if (rq > EPS) {
                    size_t off = (size_t)lane * N + best.i;
                    float avail = alloc[off] - used[off]
                        + (ext ? ext[off] : 0.f);
                    float cc = (avail + EPS) / rq;
                    if (cc > (float)BIG_CAP) cc = (float)BIG_CAP;
                    long long cl = (long long)floorf(cc);
                    ccand = max(cl, 0ll);
                }
            }
            for (int off = WAVE / 2; off > 0; off >>= 1)
                ccand = min(ccand, (long long)__shfl_down((long long)ccand,
                                                          off));
            const long long cap = __shfl((long long)ccand, 0);
            int take = (int)min(cap, (long long)remaining);
            if (take > 0) {
                if (tid == 0) {
                    ln[cursor] = best.i;
                    lc[cursor] = take;
                }
                cursor += 1;
                remaining -= take;
                if (tid < R)
                    used[(size_t)tid * N + best.i] += (float)take * req[tid];
            }
            if (!touched[best.i]) {
                // mark even on take==0 (defensive: a zero-cap winner must
                // never win again; cannot occur when snapshot caps hold)
                if (tid == 0) {
                    touched[best.i] = 1;
                    touched_list[tcount] = best.i;
                }
                tcount += 1;
            }
            __syncthreads();    // barrier 2: order stores vs next loads
        }

        int total = (int)budget - remaining;
        if (total < fuse) {
            // single-class gang revert (statement.go:375 Discard)
            for (int e = 0; e < cursor; ++e) {
                int node = ln[e];
                int cnt = lc[e];
                if (tid < R)
                    used[(size_t)tid * N + node] -= (float)cnt * req[tid];
                if (tid == 0) lc[e] = 0;
            }
            if (tid == 0) { log_len[c] = cursor; class_placed[c] = 0; }
        } else {
            if (tid < R) qa[tid] += (float)total * req[tid];
            if (tid == 0) {
                log_len[c] = cursor;
                class_placed[c] = total;
                job_placed[cd.job_idx] += total;
            }
        }
        __syncthreads();
    }
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
    unsigned* sort_scratch, int N, int R, int K, hipStream_t stream)
{
    // mega-classes (gang bundles): sort-based bulk fill — identical
    // decisions, ~16N total accesses instead of ~N per consumed node.
    // Small classes: the iterative argmax loop wins (few consumptions).
    if (sort_scratch != nullptr && ntasks >= 512 && N >= 512) {
        hipLaunchKernelGGL(vamd::bulk_select_kernel, dim3(1),
                           dim3(BS_THREADS), 0, stream, score, cap, req,
                           ntasks, used, queue_alloc, queue_limit, log_nodes,
                           log_counts, log_len, placed, job_placed, fuse_min,
                           sort_scratch, N, R, K);
        return;
    }
    hipLaunchKernelGGL(vamd::select_commit_kernel, dim3(1), dim3(SC_THREADS),
                       0, stream, score, cap, req, ntasks, used, queue_alloc,
                       queue_limit, log_nodes, log_counts, log_len, placed,
                       job_placed, fuse_min, N, R, K);
}

void vamd_fused_score_select(
    const float* alloc, float* used, const float* extra,
    const uint8_t* ready, const int64_t* taints, const int64_t* planes,
    const float* req, int64_t tolerated, const int64_t* require,
    const int64_t* forbid, float w_least, float w_most, float w_bal,
    const float* dim_w, const float* bias, float* score, int* cap,
    int ntasks, float* queue_alloc, const float* queue_limit,
    int* log_nodes, int* log_counts, int* log_len, int* placed,
    int* job_placed, int fuse_min, int N, int R, int W, int K,
    hipStream_t stream)
{
    hipLaunchKernelGGL(vamd::fused_score_select_kernel, dim3(1),
                       dim3(SC_THREADS), 0, stream, alloc, used, extra,
                       ready, taints, planes, req, tolerated, require,
                       forbid, w_least, w_most, w_bal, dim_w, bias, score,
                       cap, ntasks, queue_alloc, queue_limit, log_nodes,
                       log_counts, log_len, placed, job_placed, fuse_min,
                       N, R, W, K);
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

void vamd_batch_score(
    const VamdClassDesc* classes_dev, int c0, int count,
    const float* alloc, const float* used, const float* extra,
    const uint8_t* ready, const int64_t* taints, const int64_t* planes,
    const float* bias, const float* bias_rows,
    const float* class_req, const int64_t* class_tol_dev,
    const int64_t* class_require, const int64_t* class_forbid,
    const float* dim_w, float* score_buf,
    int N, int R, int W, hipStream_t stream)
{
    int threads = 256;
    int nx = (N + threads - 1) / threads;
    if (nx > 1024) nx = 1024;
    if (nx < 1) nx = 1;
    hipLaunchKernelGGL(vamd::batch_score_kernel, dim3(nx, count),
                       dim3(threads), 0, stream, classes_dev, c0, alloc,
                       used, extra, ready, taints, planes, bias, bias_rows,
                       class_req, class_tol_dev, class_require, class_forbid,
                       dim_w, score_buf, N, R, W);
}

void vamd_topk(float* score_buf, int count, float* topk_vals, int* topk_ids,
               int N, hipStream_t stream)
{
    hipLaunchKernelGGL(vamd::topk_kernel, dim3(count), dim3(SC_THREADS), 0,
                       stream, score_buf, topk_vals, topk_ids, N);
}

void vamd_select_chain(
    const VamdClassDesc* classes_dev, const VamdJobDesc* jobs_dev,
    int c0, int c1,
    const float* alloc, float* used, const float* extra,
    const uint8_t* ready, const int64_t* taints, const int64_t* planes,
    const float* bias, const float* bias_rows,
    const float* class_req, const int64_t* class_tol_dev,
    const int64_t* class_require, const int64_t* class_forbid,
    const float* dim_w,
    float* queue_alloc, const float* queue_limit,
    float* score_buf, const float* topk_vals, const int* topk_ids,
    int* log_nodes, int* log_counts, int* log_len,
    int* class_placed, int* job_placed,
    uint8_t* touched, int* touched_list,
    int N, int R, int W, hipStream_t stream)
{
    hipLaunchKernelGGL(vamd::select_chain_kernel, dim3(1), dim3(CH_THREADS),
                       0, stream, classes_dev, jobs_dev, c0, c1, alloc, used,
                       extra, ready, taints, planes, bias, bias_rows,
                       class_req, class_tol_dev, class_require, class_forbid,
                       dim_w, queue_alloc, queue_limit, score_buf,
                       topk_vals, topk_ids, log_nodes,
                       log_counts, log_len, class_placed, job_placed,
                       touched, touched_list, N, R, W);
}

void vamd_megacycle(
    const VamdClassDesc* classes_dev, const VamdJobDesc* jobs_dev,
    int n_jobs,
    const float* alloc, float* used, const float* extra,
    const uint8_t* ready, const int64_t* taints, const int64_t* planes,
    const float* bias, const float* bias_rows,
    const float* class_req, const int64_t* class_tol_dev,
    const int64_t* class_require, const int64_t* class_forbid,
    const int32_t* class_min, const float* dim_w,
    float* queue_alloc, const float* queue_limit,
    float* score_scratch, int* cap_scratch,
    int* log_nodes, int* log_counts, int* log_len,
    int* class_placed, int* job_placed, uint8_t* job_flag,
    int N, int R, int W, hipStream_t stream)
{
    hipLaunchKernelGGL(vamd::megacycle_kernel, dim3(1), dim3(SC_THREADS), 0,
                       stream, classes_dev, jobs_dev, n_jobs, alloc, used,
                       extra, ready, taints, planes, bias, bias_rows,
                       class_req, class_tol_dev, class_require, class_forbid,
                       class_min, dim_w, queue_alloc, queue_limit,
                       score_scratch, cap_scratch, log_nodes, log_counts,
                       log_len, class_placed, job_placed, job_flag, N, R, W);
}

}  // extern "C"
