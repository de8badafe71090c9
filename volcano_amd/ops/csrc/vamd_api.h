// C ABI of the MI355X scheduler kernel library (loaded via ctypes —
// deliberately no torch/pybind dependency: pure HIP, zero ABI hazards).
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

extern "C" {

void vamd_score_cap(
    const float* alloc, const float* used, const float* extra,
    const uint8_t* ready, const int64_t* taints, const int64_t* planes,
    const float* req, int64_t tolerated, const int64_t* require,
    const int64_t* forbid, float w_least, float w_most, float w_bal,
    const float* dim_w, const float* bias, float* score_out, int* cap_out,
    int N, int R, int W, hipStream_t stream);

void vamd_select_commit(
    float* score, const int* cap, const float* req, int ntasks, float* used,
    float* queue_alloc, const float* queue_limit, int* log_nodes,
    int* log_counts, int* log_len, int* placed, int* job_placed, int fuse_min,
    unsigned* sort_scratch,  // [4N] bulk-fill radix scratch (nullable)
    int N, int R, int K, hipStream_t stream);

void vamd_fused_score_select(
    const float* alloc, float* used, const float* extra,
    const uint8_t* ready, const int64_t* taints, const int64_t* planes,
    const float* req, int64_t tolerated, const int64_t* require,
    const int64_t* forbid, float w_least, float w_most, float w_bal,
    const float* dim_w, const float* bias, float* score, int* cap,
    int ntasks, float* queue_alloc, const float* queue_limit,
    int* log_nodes, int* log_counts, int* log_len, int* placed,
    int* job_placed, int fuse_min, int N, int R, int W, int K,
    hipStream_t stream);

void vamd_finalize_job(
    const int* job_placed, int occupied, int min_available,
    const int* class_placed, const int* class_min, uint8_t* flag, int nc,
    hipStream_t stream);

void vamd_cond_revert(
    const uint8_t* flag, int* log_nodes, int* log_counts, const int* log_len,
    const float* req, float* used, float* queue_alloc, int* placed,
    int* job_placed, int N, int R, hipStream_t stream);

// --- whole-cycle runner ----------------------------------------------------

// One task class (a batch of identical pending tasks of one job).
struct VamdClassDesc {
    int32_t job_idx;      // index into the job desc array
    int32_t queue_idx;    // row of queue_alloc / queue_limit
    int32_t ntasks;       // instances to place
    int32_t min_needed;   // per-class (role) minimum
    int32_t log_off;      // offset into the log arrays
    int32_t log_cap;      // slot capacity (>= min(ntasks, N))
    int32_t flags;        // bit0: use extra (future-idle) credit
    int32_t bias_row;     // row of bias_rows for this class, or -1 → plan bias
    float w_least, w_most, w_bal;  // score weights for this class's queue tier
    float _padf;
};

// One job (PodGroup shard-local view).
struct VamdJobDesc {
    int32_t class_begin;  // [begin, end) into the class array
    int32_t class_end;
    int32_t occupied;     // tasks already holding resources at cycle start
    int32_t min_available;
};

// Executes a whole allocate cycle: for each job (in the host-given order),
// for each of its classes: score_cap + select_commit; then the gang
// finalize/revert.  Single-class jobs fuse the gang check into
// select_commit (2 launches per job).  Everything stays on `stream`;
// NO host synchronisation happens here.
// Chain path (heterogeneous mixes): batch-score a chunk of consecutive
// single-class jobs against chunk-start usage, then replay their selects
// in one launch with lazy exact re-scoring of touched nodes.  Decisions
// are bit-identical to the per-class path (see scheduler_kernels.hip).
// Descriptor/taint arrays must be DEVICE pointers.
void vamd_batch_score(
    const VamdClassDesc* classes_dev, int c0, int count,
    const float* alloc, const float* used, const float* extra,
    const uint8_t* ready, const int64_t* taints, const int64_t* planes,
    const float* bias, const float* bias_rows,
    const float* class_req, const int64_t* class_tol_dev,
    const int64_t* class_require, const int64_t* class_forbid,
    const float* dim_w, float* score_buf,
    int N, int R, int W, hipStream_t stream);

void vamd_topk(float* score_buf, int count, float* topk_vals, int* topk_ids,
               int N, hipStream_t stream);

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
    int N, int R, int W, hipStream_t stream);

// One launch for a whole small-class plan (heterogeneous shapes):
// descriptor/taint arrays must be DEVICE pointers here.
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
    int N, int R, int W, hipStream_t stream);

void vamd_run_cycle(
    const VamdClassDesc* classes, int n_classes,
    const VamdJobDesc* jobs, int n_jobs,
    // node state [R, N] (+ masks)
    const float* alloc, float* used, const float* extra,
    const uint8_t* ready, const int64_t* taints, const int64_t* planes,
    const float* bias,
    const float* bias_rows,        // [B, N] per-class rows (nullable)
    // per-class constraint rows
    const float* class_req,        // [C, R]
    const int64_t* class_tol,      // [C]
    const int64_t* class_require,  // [C, W]
    const int64_t* class_forbid,   // [C, W]
    const int32_t* class_min,      // [C] (device copy of min_needed, for finalize)
    const float* dim_w,            // [R]
    // queue state
    float* queue_alloc,            // [Q, R]
    const float* queue_limit,      // [Q, R]
    // outputs / scratch
    float* score_scratch,          // [N]
    int* cap_scratch,              // [N]
    int* log_nodes, int* log_counts, int* log_len,  // [L], [L], [C]
    int* class_placed,             // [C]
    int* job_placed,               // [J]
    uint8_t* job_flag,             // [J]
    unsigned* sort_scratch,        // [4N] bulk-select radix scratch (nullable)
    int N, int R, int W,
    hipStream_t stream);

}  // extern "C"
