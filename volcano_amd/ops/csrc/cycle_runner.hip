// Whole-cycle launcher: replays an allocate cycle's kernel sequence from a
// host-built plan with one library call (the Python side would otherwise
// pay per-launch ctypes overhead per job; here the per-job cost is two
// hipLaunchKernelGGL enqueues).  Host code only — kernels live in
// scheduler_kernels.hip.
//
// Gang semantics (reference actions/allocate/allocate.go:719-866 +
// framework/statement.go): jobs run in the host-given priority order;
// classes of a job see each other's staged usage; a failed gang reverts
// before the next job's first kernel, so the sequential-cycle semantics
// are preserved exactly while the whole cycle stays on-device.

#include "vamd_api.h"

#include <cstdlib>
#include <cstring>

// one launch for the whole plan when every class is small.  Measured
// OFF by default: at 10k nodes the async per-class enqueue pipeline
// overlaps launches with host plan iteration, while the single-
// workgroup megacycle serializes the per-class score passes (mix bench
// 559 ms vs 778 ms/step — gpurun_out/g28_mix*.log).  Kept as an
// opt-in (VAMD_MEGACYCLE=1) for launch-latency-bound deployments
// (small N, many classes); a cooperative multi-block variant is the
// roadmap item.  getenv per cycle so tests can flip it.
static bool megacycle_enabled() {
    const char* e = getenv("VAMD_MEGACYCLE");
    return e != nullptr && atoi(e) != 0;
}

extern "C" void vamd_run_cycle(
    const VamdClassDesc* classes, int n_classes,
    const VamdJobDesc* jobs, int n_jobs,
    const float* alloc, float* used, const float* extra,
    const uint8_t* ready, const int64_t* taints, const int64_t* planes,
    const float* bias,
    const float* bias_rows,
    const float* class_req, const int64_t* class_tol,
    const int64_t* class_require, const int64_t* class_forbid,
    const int32_t* class_min, const float* dim_w,
    float* queue_alloc, const float* queue_limit,
    float* score_scratch, int* cap_scratch,
    int* log_nodes, int* log_counts, int* log_len,
    int* class_placed, int* job_placed, uint8_t* job_flag,
    unsigned* sort_scratch,
    int N, int R, int W, hipStream_t stream)
{
    bool any_bulk = false;
    for (int c = 0; c < n_classes; ++c)
        if (classes[c].ntasks >= 512) { any_bulk = true; break; }
    if (megacycle_enabled() && !any_bulk && n_classes >= 32) {
        // stage descriptors + taint masks on device (stream-ordered)
        size_t sz_c = (size_t)n_classes * sizeof(VamdClassDesc);
        size_t sz_j = (size_t)n_jobs * sizeof(VamdJobDesc);
        size_t sz_t = (size_t)n_classes * sizeof(int64_t);
        void* dev_blob = nullptr;
        if (hipMallocAsync(&dev_blob, sz_c + sz_j + sz_t, stream)
                == hipSuccess && dev_blob != nullptr) {
            char* p = (char*)dev_blob;
            (void)hipMemcpyAsync(p, classes, sz_c, hipMemcpyHostToDevice,
                                 stream);
            (void)hipMemcpyAsync(p + sz_c, jobs, sz_j,
                                 hipMemcpyHostToDevice, stream);
            (void)hipMemcpyAsync(p + sz_c + sz_j, class_tol, sz_t,
                                 hipMemcpyHostToDevice, stream);
            vamd_megacycle((const VamdClassDesc*)p,
                           (const VamdJobDesc*)(p + sz_c), n_jobs,
                           alloc, used, extra, ready, taints, planes,
                           bias, bias_rows, class_req,
                           (const int64_t*)(p + sz_c + sz_j),
                           class_require, class_forbid, class_min,
                           dim_w, queue_alloc, queue_limit,
                           score_scratch, cap_scratch, log_nodes,
                           log_counts, log_len, class_placed,
                           job_placed, job_flag, N, R, W, stream);
            (void)hipFreeAsync(dev_blob, stream);
            return;
        }
        // allocation failed: fall through to per-class launches
    }

    // ---- chain path: runs of consecutive single-class small jobs become
    // 2 launches per chunk (batch score + select chain) instead of 2 per
    // class.  Decisions match the per-class path exactly (lazy re-score
    // of touched nodes — scheduler_kernels.hip).  VAMD_NO_CHAIN=1 kills it.
    const int CHAIN_MIN = 32;
    int CHUNK_MAX = 512;
    {
        const char* e = getenv("VAMD_CHAIN_CHUNK");
        if (e != nullptr && atoi(e) > 0) CHUNK_MAX = atoi(e);
    }
    auto chainable = [&](int j) {
        const VamdJobDesc& jb = jobs[j];
        if (jb.class_end - jb.class_begin != 1) return false;
        return classes[jb.class_begin].ntasks < 512;
    };
    bool chain_on = false;
    {
        const char* e = getenv("VAMD_NO_CHAIN");
        bool disabled = e != nullptr && atoi(e) != 0;
        // Measured OFF by default (like the megacycle): the batched
        // score + topk phases are cheap, but the serial select chain
        // degrades under placement CONCENTRATION — when every class
        // prefers the same nodes, the top-K candidate lists exhaust and
        // each pop pays a full-row fallback scan (mix bench: 199-291 ms
        // vs 157 ms per-class at 10k nodes; 1.9 s vs 0.7 s at 50k —
        // gpurun_out/bench_r2e-g*).  The async per-class pipeline keeps
        // the GPU busy at ~12.5 us/class.  VAMD_CHAIN_CHUNK=<n> opts in
        // (the equivalence test exercises it; decisions are exact).
        if (getenv("VAMD_CHAIN_CHUNK") == nullptr)
            disabled = true;
        if (!disabled) {
            int run = 0;
            for (int j = 0; j < n_jobs; ++j) {
                bool ok = chainable(j) &&
                    (run == 0 || jobs[j].class_begin == jobs[j - 1].class_end);
                run = ok ? run + 1 : (chainable(j) ? 1 : 0);
                if (run >= CHAIN_MIN) { chain_on = true; break; }
            }
        }
    }
    char* chain_blob = nullptr;
    const VamdClassDesc* cdev = nullptr;
    const VamdJobDesc* jdev = nullptr;
    const int64_t* tdev = nullptr;
    float* score_buf = nullptr;
    int* tlist = nullptr;
    float* topk_vals = nullptr;
    int* topk_ids = nullptr;
    uint8_t* touched = nullptr;
    if (chain_on) {
        size_t per_class = (size_t)N * sizeof(float);
        size_t mem_cap = (size_t)256 << 20;
        if ((size_t)CHUNK_MAX * per_class > mem_cap)
            CHUNK_MAX = (int)(mem_cap / per_class);
        if (CHUNK_MAX < 8) CHUNK_MAX = 8;
        size_t sz_c = (size_t)n_classes * sizeof(VamdClassDesc);
        size_t sz_j = (size_t)n_jobs * sizeof(VamdJobDesc);
        size_t sz_t = (size_t)n_classes * sizeof(int64_t);
        size_t sz_s = (size_t)CHUNK_MAX * per_class;
        size_t sz_l = (size_t)N * sizeof(int);
        size_t sz_b = (size_t)N;
        size_t sz_tv = (size_t)CHUNK_MAX * 16 * sizeof(float);
        size_t sz_ti = (size_t)CHUNK_MAX * 16 * sizeof(int);
        void* p = nullptr;
        if (hipMallocAsync(&p, sz_c + sz_j + sz_t + sz_s + sz_l + sz_b
                           + sz_tv + sz_ti,
                           stream) == hipSuccess && p != nullptr) {
            chain_blob = (char*)p;
            char* q = chain_blob;
            (void)hipMemcpyAsync(q, classes, sz_c, hipMemcpyHostToDevice,
                                 stream);
            cdev = (const VamdClassDesc*)q; q += sz_c;
            (void)hipMemcpyAsync(q, jobs, sz_j, hipMemcpyHostToDevice,
                                 stream);
            jdev = (const VamdJobDesc*)q; q += sz_j;
            (void)hipMemcpyAsync(q, class_tol, sz_t, hipMemcpyHostToDevice,
                                 stream);
            tdev = (const int64_t*)q; q += sz_t;
            score_buf = (float*)q; q += sz_s;
            tlist = (int*)q; q += sz_l;
            topk_vals = (float*)q; q += sz_tv;
            topk_ids = (int*)q; q += sz_ti;
            touched = (uint8_t*)q;
        } else {
            chain_on = false;
        }
    }

    for (int j = 0; j < n_jobs; ) {
        if (chain_on && chainable(j)) {
            // maximal contiguous chain run from j
            int e = j + 1;
            while (e < n_jobs && chainable(e)
                   && jobs[e].class_begin == jobs[e - 1].class_end)
                ++e;
            if (e - j >= CHAIN_MIN) {
                int cb = jobs[j].class_begin;
                int ce = jobs[e - 1].class_end;
                for (int cc = cb; cc < ce; cc += CHUNK_MAX) {
                    int cend = cc + CHUNK_MAX < ce ? cc + CHUNK_MAX : ce;
                    vamd_batch_score(cdev, cc, cend - cc, alloc, used, extra,
                                     ready, taints, planes, bias, bias_rows,
                                     class_req, tdev, class_require,
                                     class_forbid, dim_w, score_buf,
                                     N, R, W, stream);
                    vamd_topk(score_buf, cend - cc, topk_vals, topk_ids, N,
                              stream);
                    vamd_select_chain(cdev, jdev, cc, cend, alloc, used,
                                      extra, ready, taints, planes, bias,
                                      bias_rows, class_req, tdev,
                                      class_require, class_forbid, dim_w,
                                      queue_alloc, queue_limit, score_buf,
                                      topk_vals, topk_ids,
                                      log_nodes, log_counts, log_len,
                                      class_placed, job_placed, touched,
                                      tlist, N, R, W, stream);
                }
                j = e;
                continue;
            }
        }
        const VamdJobDesc& job = jobs[j];
        int nc = job.class_end - job.class_begin;
        bool single = (nc == 1);

        for (int c = job.class_begin; c < job.class_end; ++c) {
            const VamdClassDesc& cd = classes[c];
            const float* ext = (cd.flags & 1) ? extra : nullptr;
            // per-class bias row (task-topology bucket packing) overrides
            // the plan-wide bias plane
            const float* b = (cd.bias_row >= 0 && bias_rows)
                ? bias_rows + (size_t)cd.bias_row * N : bias;

            // single-class jobs: gang check fused into the commit.  The
            // effective minimum is how many more tasks the job needs to
            // become ready (min_available - occupied), but never less than
            // the class's own role minimum.
            int fuse_min = -1;
            if (single) {
                int need = job.min_available - job.occupied;
                if (cd.min_needed > need) need = cd.min_needed;
                fuse_min = need > 0 ? need : 0;
            }

            if (cd.ntasks < 512 && N <= 4096) {
                // small class on a small inventory: ONE fused launch
                // (score in-block + select).  At larger N the one-block
                // score loses the multi-CU latency hiding of the
                // grid-stride kernel (mix bench at N=10k measured
                // 552 ms/step fused vs 334 ms split).
                vamd_fused_score_select(
                    alloc, used, ext, ready, taints, planes,
                    class_req + (size_t)c * R, class_tol[c],
                    class_require + (size_t)c * W,
                    class_forbid + (size_t)c * W,
                    cd.w_least, cd.w_most, cd.w_bal, dim_w, b,
                    score_scratch, cap_scratch, cd.ntasks,
                    queue_alloc + (size_t)cd.queue_idx * R,
                    queue_limit + (size_t)cd.queue_idx * R,
                    log_nodes + cd.log_off, log_counts + cd.log_off,
                    log_len + c, class_placed + c, job_placed + j,
                    fuse_min, N, R, W, cd.log_cap, stream);
                continue;
            }

            vamd_score_cap(alloc, used, ext, ready, taints, planes,
                           class_req + (size_t)c * R, class_tol[c],
                           class_require + (size_t)c * W,
                           class_forbid + (size_t)c * W,
                           cd.w_least, cd.w_most, cd.w_bal, dim_w, b,
                           score_scratch, cap_scratch, N, R, W, stream);

            vamd_select_commit(score_scratch, cap_scratch,
                               class_req + (size_t)c * R, cd.ntasks, used,
                               queue_alloc + (size_t)cd.queue_idx * R,
                               queue_limit + (size_t)cd.queue_idx * R,
                               log_nodes + cd.log_off, log_counts + cd.log_off,
                               log_len + c, class_placed + c, job_placed + j,
                               fuse_min, sort_scratch, N, R, cd.log_cap,
                               stream);
        }

        if (!single) {
            vamd_finalize_job(job_placed + j, job.occupied, job.min_available,
                              class_placed + job.class_begin,
                              class_min + job.class_begin, job_flag + j, nc,
                              stream);
            for (int c = job.class_begin; c < job.class_end; ++c) {
                const VamdClassDesc& cd = classes[c];
                vamd_cond_revert(job_flag + j, log_nodes + cd.log_off,
                                 log_counts + cd.log_off, log_len + c,
                                 class_req + (size_t)c * R, used,
                                 queue_alloc + (size_t)cd.queue_idx * R,
                                 class_placed + c, job_placed + j, N, R,
                                 stream);
            }
        }
        ++j;
    }
    if (chain_blob != nullptr)
        (void)hipFreeAsync(chain_blob, stream);
}
