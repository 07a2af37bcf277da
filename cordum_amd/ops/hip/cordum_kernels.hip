// CDNA4 (gfx950) kernels for the cordum_amd control plane.
//
// K1 policy_first_match   — batched rule×job policy evaluation
//                           (replaces SafetyPolicy.Evaluate, safety_policy.go:187-294)
// K2 least_loaded_pick    — batched worker scoring + masked argmin
//                           (replaces PickSubject, strategy_least_loaded.go:40-136)
// K4 deadline_scan        — deadline/timeout sweep over the job table
//                           (replaces reconciler.go:88-144 + job_store deadline ZSET)
// K5 apply_transitions    — batched job state transitions with the legality LUT
//                           (replaces job_store.go:249-329 WATCH/tx)
// W  echo_worker          — device worker pool execution (payload touch + result)
//
// Design notes (see /opt/skills/guides/cdna_hip_programming.md):
//  - wave64; blocks of 256 threads; grid-stride where applicable.
//  - policy kernel: one job per thread, rules staged through LDS in chunks by
//    the whole workgroup (rule rows are reused by all 256 threads; LDS
//    staging turns R global reads per thread into R/256 per thread).
//  - first-match across rule chunks uses atomicMin on the output word (the
//    output is initialized to INT_MAX by the host); chunk-level early-out
//    via a workgroup ballot on "all jobs in this block already matched an
//    earlier rule".
//  - scoring kernel: score = active + cpu/100 + gpu/100 packed into a
//    monotonic (score,idx) uint64 key -> deterministic argmin (ties by
//    lowest worker index, matching the host strategy).
#include <hip/hip_runtime.h>
#include <cstdint>

#define WAVE 64
#define BLOCK 256

// ---------------------------------------------------------------------------
// K1: policy first-match
// ---------------------------------------------------------------------------
// Rule row layout (int64 words, W words per dim):
//   any[7*W] | all[2*W] | mcp_allow[4*W] | mcp_deny[4*W]
// plus per-rule bytes: secrets(int8), mcp_any(u8).
// Job row: any[7*W] | all[2*W] | mcp[4*W], secrets u8, mcp_used u8.

template <int W, int JPT>
__global__ __launch_bounds__(BLOCK) void policy_first_match_kernel(
    const long long* __restrict__ rule_any,   // [R,7,W]
    const long long* __restrict__ rule_all,   // [R,2,W]
    const signed char* __restrict__ rule_secrets, // [R]
    const long long* __restrict__ rule_mcp,   // [R,4,2,W]
    const unsigned char* __restrict__ rule_mcp_any, // [R]
    const long long* __restrict__ job_any,    // [J,7,W]
    const long long* __restrict__ job_all,    // [J,2,W]
    const unsigned char* __restrict__ job_secrets, // [J]
    const long long* __restrict__ job_mcp,    // [J,4,W]
    const unsigned char* __restrict__ job_mcp_used, // [J]
    int* __restrict__ out_first,              // [J], pre-filled INT_MAX
    int R, int J, int rules_per_chunk)
{
    // 2D grid: blockIdx.x tiles jobs (JPT jobs per thread — the rule stream
    // is the traffic bound at large R, and every extra job per block divides
    // it), blockIdx.y tiles the rule range; first-match semantics restored
    // by the atomicMin on the output word.
    const int rule_chunk_begin = blockIdx.y * rules_per_chunk;
    const int rule_chunk_end = min(R, rule_chunk_begin + rules_per_chunk);
    constexpr int ANY_W = 7 * W;
    constexpr int ALL_W = 2 * W;
    constexpr int MCP_W = 4 * W;
    constexpr int ROW = ANY_W + ALL_W + 2 * MCP_W;     // int64 words per rule
    constexpr int CHUNK = 96;                          // rules staged per LDS pass

    __shared__ long long lds_rules[CHUNK * ROW];
    __shared__ signed char lds_secrets[CHUNK];
    __shared__ unsigned char lds_mcp_any[CHUNK];
    __shared__ int block_done;

    int jj[JPT];
    long long jany[JPT][ANY_W], jall[JPT][ALL_W], jmcp[JPT][MCP_W];
    unsigned char jsec[JPT], jused[JPT];
    int best[JPT];
    #pragma unroll
    for (int k = 0; k < JPT; ++k) {
        const int j = (blockIdx.x * JPT + k) * BLOCK + threadIdx.x;
        jj[k] = j;
        best[k] = INT_MAX;
        if (j < J) {
            #pragma unroll
            for (int w = 0; w < ANY_W; ++w) jany[k][w] = job_any[(size_t)j * ANY_W + w];
            #pragma unroll
            for (int w = 0; w < ALL_W; ++w) jall[k][w] = job_all[(size_t)j * ALL_W + w];
            #pragma unroll
            for (int w = 0; w < MCP_W; ++w) jmcp[k][w] = job_mcp[(size_t)j * MCP_W + w];
            jsec[k] = job_secrets[j];
            jused[k] = job_mcp_used[j];
        } else {
            best[k] = INT_MAX - 1;  // dead
        }
    }

    for (int base = rule_chunk_begin; base < rule_chunk_end; base += CHUNK) {
        const int n = min(CHUNK, rule_chunk_end - base);
        for (int i = threadIdx.x; i < n * ROW; i += BLOCK) {
            const int r = i / ROW, w = i % ROW;
            const size_t g = (size_t)(base + r);
            long long v;
            if (w < ANY_W)                 v = rule_any[g * ANY_W + w];
            else if (w < ANY_W + ALL_W)    v = rule_all[g * ALL_W + (w - ANY_W)];
            else                           v = rule_mcp[g * 2 * MCP_W + (w - ANY_W - ALL_W)];
            lds_rules[i] = v;
        }
        for (int i = threadIdx.x; i < n; i += BLOCK) {
            lds_secrets[i] = rule_secrets[base + i];
            lds_mcp_any[i] = rule_mcp_any[base + i];
        }
        __syncthreads();

        #pragma unroll
        for (int k = 0; k < JPT; ++k) {
            // cross-chunk early-skip (atomicMin output is monotonic; a stale
            // read only wastes a scan, never correctness)
            if (best[k] == INT_MAX && base > 0 &&
                __hip_atomic_load(&out_first[jj[k]], __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT) < base) {
                best[k] = INT_MAX - 1;
            }
            if (best[k] != INT_MAX) continue;
            for (int r = 0; r < n; ++r) {
                const long long* row = &lds_rules[r * ROW];
                bool ok = true;
                #pragma unroll
                for (int d = 0; d < 7 && ok; ++d) {
                    long long inter = 0, any = 0;
                    #pragma unroll
                    for (int w = 0; w < W; ++w) {
                        const long long rm = row[d * W + w];
                        any |= rm;
                        inter |= rm & jany[k][d * W + w];
                    }
                    ok = (any == 0) | (inter != 0);
                }
                #pragma unroll
                for (int d = 0; d < 2 && ok; ++d) {
                    long long missing = 0;
                    #pragma unroll
                    for (int w = 0; w < W; ++w)
                        missing |= row[ANY_W + d * W + w] & ~jall[k][d * W + w];
                    ok = (missing == 0);
                }
                if (ok) {
                    const signed char rs = lds_secrets[r];
                    ok = (rs < 0) | (rs == (signed char)jsec[k]);
                }
                if (ok && jused[k] && lds_mcp_any[r]) {
                    const long long* allow = &row[ANY_W + ALL_W];
                    const long long* deny = &row[ANY_W + ALL_W + MCP_W];
                    #pragma unroll
                    for (int f = 0; f < 4 && ok; ++f) {
                        long long a = 0, ainter = 0, dinter = 0;
                        #pragma unroll
                        for (int w = 0; w < W; ++w) {
                            a |= allow[f * W + w];
                            ainter |= allow[f * W + w] & jmcp[k][f * W + w];
                            dinter |= deny[f * W + w] & jmcp[k][f * W + w];
                        }
                        ok = (dinter == 0) & ((a == 0) | (ainter != 0));
                    }
                }
                if (ok) { best[k] = base + r; break; }
            }
        }
        __syncthreads();
        // chunk early-out: every live job in this block already matched
        if (threadIdx.x == 0) block_done = 1;
        __syncthreads();
        #pragma unroll
        for (int k = 0; k < JPT; ++k)
            if (best[k] == INT_MAX) block_done = 0;
        __syncthreads();
        if (block_done) break;
    }

    #pragma unroll
    for (int k = 0; k < JPT; ++k)
        if (jj[k] < J && best[k] < INT_MAX - 1) atomicMin(&out_first[jj[k]], best[k]);
}

// ---------------------------------------------------------------------------
// K2: least-loaded worker pick
// ---------------------------------------------------------------------------
// Worker row: pool(int32), active(int32), maxp(int32), cpu(f32), gpu(f32),
// labels_mask(int64). Job row: pool_mask(int64 over pool vocab), required
// labels mask(int64), preferred_worker(int32, -1 none).
// Output per job: packed pick int32 (-1 no_workers, -2 overloaded) .

__device__ __forceinline__ unsigned long long score_key(float score, int widx) {
    // score >= 0 -> float bits are monotonic; tie-break by lowest index
    unsigned int sb = __float_as_uint(score);
    return ((unsigned long long)sb << 32) | (unsigned int)widx;
}

__global__ __launch_bounds__(BLOCK) void worker_precompute_kernel(
    const int* __restrict__ w_pool,        // [W]
    const int* __restrict__ w_active,      // [W]
    const int* __restrict__ w_maxp,        // [W]
    const float* __restrict__ w_cpu,       // [W]
    const float* __restrict__ w_gpu,       // [W]
    unsigned long long* __restrict__ w_key, // [W] out: (score,idx) or flags
    int NW)
{
    // per-worker tick-invariant precompute: pack (score, idx) into a
    // monotonic u64 key; overloaded workers get OVERLOADED_KEY so the picker
    // can still distinguish "all overloaded" from "no workers".
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    if (i >= NW) return;
    const int active = w_active[i], maxp = w_maxp[i];
    const float cpu = w_cpu[i], gpu = w_gpu[i];
    bool over = false;
    if (maxp > 0 && (float)active / (float)maxp >= 0.9f) over = true;
    if (cpu >= 90.f || gpu >= 90.f) over = true;
    if (over) {
        w_key[i] = 0xfffffffe00000000ull | (unsigned int)i;  // OVERLOADED
    } else {
        const float score = (float)active + cpu * 0.01f + gpu * 0.01f;
        w_key[i] = score_key(score, i);
    }
}

#define OVERLOADED_TAG 0xfffffffeu

__global__ __launch_bounds__(BLOCK) void least_loaded_pick_kernel(
    const int* __restrict__ w_pool,         // [W]
    const unsigned long long* __restrict__ w_key, // [W] from precompute
    const long long* __restrict__ w_labels, // [W]
    const long long* __restrict__ j_poolmask, // [J]
    const long long* __restrict__ j_labels,   // [J] required labels
    int* __restrict__ out_pick,             // [J]
    int NW, int NJ,
    const int* __restrict__ valid_count,    // [1] or nullptr: skip gate
    long long full_mask)
{
    // K2c short-circuit: when any worker is non-overloaded, unconstrained
    // jobs take the spread (round-robin over the sorted order) in the
    // compaction kernel — their exact scan pick is dead work. The full
    // 1000-worker scan was 15.5% of the tick for picks that were
    // overwritten (profiles/r15_bench_ktrace_stats.txt). -1 here matches
    // the all-overloaded scan outcome (-2): both are "no direct route".
    // (block-uniform check: the scan below has __syncthreads() barriers
    // shared by the block's 4 waves, so only a whole-block skip is legal)
    if (valid_count != nullptr && *valid_count > 0) {
        const int jb = blockIdx.x * (BLOCK / WAVE);
        bool all_skip = true;
        for (int k = 0; k < BLOCK / WAVE; ++k) {
            const int jj = jb + k;
            if (jj < NJ && (j_labels[jj] != 0 || j_poolmask[jj] != full_mask)) {
                all_skip = false;
                break;
            }
        }
        if (all_skip) {
            const int jj = jb + threadIdx.x / WAVE;
            if (jj < NJ && threadIdx.x % WAVE == 0) out_pick[jj] = -1;
            return;
        }
    }
    // one WAVE per job: 64 lanes stride the worker table (staged through LDS
    // once per workgroup, shared by the 4 waves), then a wave min-reduce.
    constexpr int WCHUNK = 1024;
    __shared__ unsigned long long s_key[WCHUNK];
    __shared__ long long s_labels[WCHUNK];
    __shared__ int s_pool[WCHUNK];

    const int wave = threadIdx.x / WAVE;           // 0..3
    const int lane = threadIdx.x % WAVE;
    const int j = blockIdx.x * (BLOCK / WAVE) + wave;

    long long pool_mask = 0, req_labels = 0;
    if (j < NJ) {
        pool_mask = j_poolmask[j];
        req_labels = j_labels[j];
    }

    unsigned long long best = ~0ull;
    int overloaded = 0, total = 0;

    for (int base = 0; base < NW; base += WCHUNK) {
        const int n = min(WCHUNK, NW - base);
        for (int i = threadIdx.x; i < n; i += BLOCK) {
            s_key[i] = w_key[base + i];
            s_labels[i] = w_labels[base + i];
            s_pool[i] = w_pool[base + i];
        }
        __syncthreads();
        if (j < NJ) {
            for (int i = lane; i < n; i += WAVE) {
                const int pool = s_pool[i];
                if (pool < 0 || pool >= 64) continue;
                if (!((pool_mask >> pool) & 1)) continue;
                if (req_labels & ~s_labels[i]) continue;
                ++total;
                const unsigned long long key = s_key[i];
                if ((unsigned int)(key >> 32) == OVERLOADED_TAG) { ++overloaded; continue; }
                if (key < best) best = key;
            }
        }
        __syncthreads();
    }

    // wave reductions (64-lane shuffles)
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
        const unsigned long long other = __shfl_down(best, off, WAVE);
        if (other < best) best = other;
        overloaded += __shfl_down(overloaded, off, WAVE);
        total += __shfl_down(total, off, WAVE);
    }
    if (lane == 0 && j < NJ) {
        if (best != ~0ull) out_pick[j] = (int)(best & 0xffffffffu);
        else if (total > 0 && overloaded == total) out_pick[j] = -2;
        else out_pick[j] = -1;
    }
}

// K2c: batch spreading — unconstrained jobs round-robin over the K least
// loaded workers. A frozen-snapshot argmin sends a whole homogeneous batch
// to ONE worker (the reference has the same pathology between heartbeats:
// its scheduler replica piles onto the lowest scorer until the next 10 s
// heartbeat); queue-group fan-in is the intended behavior, and spreading a
// batch across the least-loaded set IS that behavior for a batched tick.
// Constrained jobs (labels / narrowed pool mask) keep their exact scan pick.
__global__ __launch_bounds__(BLOCK) void spread_pick_kernel(
    const int* __restrict__ order,        // [NW] worker idx sorted by key asc
    const int* __restrict__ valid_count,  // [1] non-overloaded workers
    const long long* __restrict__ j_poolmask,
    const long long* __restrict__ j_labels,
    long long full_mask,
    int* __restrict__ pick,               // in-out
    int NJ, int K)
{
    const int j = blockIdx.x * BLOCK + threadIdx.x;
    if (j >= NJ) return;
    if (j_labels[j] != 0 || j_poolmask[j] != full_mask) return;  // constrained
    const int V = min(*valid_count, K);
    if (V > 0) pick[j] = order[j % V];
}

// ---------------------------------------------------------------------------
// K5: batched state transitions with legality LUT (job_store.go:70-82)
// ---------------------------------------------------------------------------
#define N_STATES 11
__constant__ unsigned char d_transition_lut[N_STATES * N_STATES];

__global__ __launch_bounds__(BLOCK) void apply_transitions_kernel(
    unsigned char* __restrict__ states,     // [N] job table states
    int* __restrict__ attempts,             // [N]
    const int* __restrict__ slots,          // [B] job slots to transition
    const unsigned char* __restrict__ to_states, // [B] target states
    unsigned char* __restrict__ ok_out,     // [B] 1 = applied
    long long* __restrict__ deadlines,      // [N] cleared on terminal
    int B)
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    if (i >= B) return;
    const int slot = slots[i];
    const unsigned char to = to_states[i];
    const unsigned char from = states[slot];
    const unsigned char legal = d_transition_lut[from * N_STATES + to];
    ok_out[i] = legal;
    if (!legal) return;
    states[slot] = to;
    // attempts increment on entering SCHEDULED (JobState.SCHEDULED == 3)
    if (to == 3 && from != 3) attempts[slot] += 1;
    // terminal states clear the deadline (SUCCEEDED..DENIED are 6..10)
    if (to >= 6) deadlines[slot] = (long long)0x7fffffffffffffffLL;
}

// wave-aggregated compaction append: one atomicAdd per wave instead of one
// per lane (a single counter word saturates at ~88 atomics/us on this chip —
// MI355X_MICROARCH.md price-list row "dequeue")
__device__ __forceinline__ int wave_append_slot(bool pred, int* counter, int lane) {
    const unsigned long long mask = __ballot(pred);
    const int nactive = __popcll(mask);
    int base = 0;
    const int leader = __ffsll((long long)mask) - 1;
    if (pred && lane == leader) base = atomicAdd(counter, nactive);
    base = __shfl(base, leader, WAVE);
    const int rank = __popcll(mask & ((1ull << lane) - 1ull));
    return pred ? base + rank : -1;
}

// K7: DLQ ring append — denied/failed slots into a capped device ring
// (dlq_store.go's capped index as an HBM ring with an atomic head)
__global__ __launch_bounds__(BLOCK) void dlq_ring_append_kernel(
    const int* __restrict__ slots,     // [<=cap]
    const int* __restrict__ count,     // [1]
    int* __restrict__ ring,            // [ring_size]
    int* __restrict__ head,            // [1] monotonically increasing
    int ring_size)
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    const int lane = threadIdx.x % WAVE;
    const bool live = i < *count;
    const int pos = wave_append_slot(live, head, lane);
    if (pos >= 0) ring[pos % ring_size] = live ? slots[i] : -1;
}


// ---------------------------------------------------------------------------
// K3: run/step readiness sweep over HBM-resident run state
// ---------------------------------------------------------------------------
// Batched equivalent of scheduleReady's dependency gate (workflow/engine.go
// :453-827, depsSatisfied :1231-1242) for runs with <= 64 steps: per run,
// step s is dispatchable when
//   step_state[s] == PENDING(0)  AND  deps_mask[s] subset-of succeeded_mask
// where succeeded_mask bit t = (step_state[t] == SUCCEEDED). Emits a
// compacted (run, step) dispatch list. One thread per run; 64-step bitmask
// per run row (larger workflows stay on the host path).
#define STEP_PENDING_C 0
#define STEP_SUCCEEDED_C 3

__global__ __launch_bounds__(BLOCK) void run_readiness_kernel(
    const unsigned char* __restrict__ step_state, // [NR][64]
    const long long* __restrict__ deps_mask,      // [NR][64]
    const unsigned char* __restrict__ n_steps,    // [NR]
    const unsigned char* __restrict__ run_active, // [NR] 1 = pending/running
    long long* __restrict__ ready_mask_out,       // [NR]
    int* __restrict__ dispatch_runs,              // [cap]
    int* __restrict__ dispatch_steps,             // [cap]
    int* __restrict__ dispatch_count,             // [1]
    int NR, int cap)
{
    const int run = blockIdx.x * BLOCK + threadIdx.x;
    const int lane = threadIdx.x % WAVE;
    long long ready = 0;
    int ns = 0;
    if (run < NR && run_active[run]) {
        ns = n_steps[run];
        const unsigned char* st = &step_state[(size_t)run * 64];
        unsigned long long succeeded = 0;
        #pragma unroll 8
        for (int t = 0; t < ns; ++t)
            succeeded |= (unsigned long long)(st[t] == STEP_SUCCEEDED_C) << t;
        const long long* dm = &deps_mask[(size_t)run * 64];
        for (int s = 0; s < ns; ++s) {
            if (st[s] != STEP_PENDING_C) continue;
            const unsigned long long need = (unsigned long long)dm[s];
            if ((need & ~succeeded) == 0) ready |= 1ll << s;
        }
    }
    if (run < NR) ready_mask_out[run] = ready;
    // compact the dispatch list, one wave-aggregated append per ready step
    while (true) {
        const bool has = ready != 0;
        if (!__any(has)) break;
        int step = -1;
        if (has) {
            step = __ffsll(ready) - 1;
            ready &= ready - 1;
        }
        const int pos = wave_append_slot(step >= 0, dispatch_count, lane);
        if (pos >= 0 && pos < cap) {
            dispatch_runs[pos] = run;
            dispatch_steps[pos] = step;
        }
    }
}

// ---------------------------------------------------------------------------
// K3-WF: device-resident workflow engine tick (config #3)
// ---------------------------------------------------------------------------
// Batched equivalent of the per-run scheduleReady walk + HandleJobResult +
// updateRunStatus loop (workflow/engine.go:453-827, :1524-1560, :1623-1699)
// over HBM-resident run tables. Step kinds: WORKER (1 child job), FOR_EACH
// (fanout_n children, aggregated like engine.go:1623-1645), APPROVAL
// (WAITING hold until a host grant — the human-in-the-loop hop stays on the
// host, like gateway.go:3553), CONDITION (pre-evaluated bit -> SUCCEEDED or
// SKIPPED), DELAY (next_ready tick gate). Failed children retry with
// exponential tick backoff up to max_retries (computeBackoff analog), then
// the step FAILs and the run FAILs (failure blocks downstream,
// depsSatisfied :1231-1242 — SKIPPED satisfies, FAILED never does).
#define WFS_PENDING 0
#define WFS_DISPATCHED 1
#define WFS_WAITING 2
#define WFS_SUCCEEDED 3
#define WFS_FAILED 4
#define WFS_SKIPPED 5
#define WFK_WORKER 0
#define WFK_FOR_EACH 1
#define WFK_APPROVAL 2
#define WFK_CONDITION 3
#define WFK_DELAY 4

__global__ __launch_bounds__(BLOCK) void wf_sweep_kernel(
    unsigned char* __restrict__ step_state,   // [NR*64]
    const long long* __restrict__ deps_mask,  // [NR*64]
    const unsigned char* __restrict__ n_steps,// [NR]
    const unsigned char* __restrict__ run_active, // [NR]
    const unsigned char* __restrict__ step_kind,  // [NR*64]
    const long long* __restrict__ cond_bits,  // [NR] bit s = condition value
    const int* __restrict__ next_ready,       // [NR*64] tick gate
    const int* __restrict__ tick_p,           // [1] device tick counter
    int* __restrict__ disp_runs,              // [cap] WORKER/FOR_EACH to expand
    int* __restrict__ disp_steps,             // [cap]
    int* __restrict__ disp_count,             // [1]
    int* __restrict__ appr_runs,              // [acap] fresh WAITING approvals
    int* __restrict__ appr_steps,             // [acap]
    int* __restrict__ appr_count,             // [1]
    int NR, int cap, int acap)
{
    const int run = blockIdx.x * BLOCK + threadIdx.x;
    const int lane = threadIdx.x % WAVE;
    const int tick = *tick_p;
    long long disp = 0, appr = 0;
    if (run < NR && run_active[run]) {
        const int ns = n_steps[run];
        unsigned char* st = &step_state[(size_t)run * 64];
        unsigned long long satisfied = 0;
        for (int t = 0; t < ns; ++t) {
            const unsigned char s = st[t];
            satisfied |= (unsigned long long)(s == WFS_SUCCEEDED || s == WFS_SKIPPED) << t;
        }
        const long long* dm = &deps_mask[(size_t)run * 64];
        const long long cb = cond_bits[run];
        for (int s = 0; s < ns; ++s) {
            if (st[s] != WFS_PENDING) continue;
            if (((unsigned long long)dm[s] & ~satisfied) != 0) continue;
            if (next_ready[(size_t)run * 64 + s] > tick) continue;
            switch (step_kind[(size_t)run * 64 + s]) {
            case WFK_WORKER:
            case WFK_FOR_EACH:
                disp |= 1ll << s;
                break;
            case WFK_APPROVAL:
                st[s] = WFS_WAITING;   // held for the host-side grant
                appr |= 1ll << s;
                break;
            case WFK_CONDITION:
                st[s] = ((cb >> s) & 1) ? WFS_SUCCEEDED : WFS_SKIPPED;
                break;
            case WFK_DELAY:            // gate already passed
                st[s] = WFS_SUCCEEDED;
                break;
            }
        }
    }
    while (true) {   // wave-aggregated appends, one ready step per pass
        if (!__any(disp != 0)) break;
        int step = -1;
        if (disp != 0) { step = __ffsll(disp) - 1; disp &= disp - 1; }
        const int pos = wave_append_slot(step >= 0, disp_count, lane);
        if (pos >= 0 && pos < cap) { disp_runs[pos] = run; disp_steps[pos] = step; }
    }
    while (true) {
        if (!__any(appr != 0)) break;
        int step = -1;
        if (appr != 0) { step = __ffsll(appr) - 1; appr &= appr - 1; }
        const int pos = wave_append_slot(step >= 0, appr_count, lane);
        if (pos >= 0 && pos < acap) { appr_runs[pos] = run; appr_steps[pos] = step; }
    }
}

// for_each expansion: one wave per dispatch entry; all-or-nothing child-slot
// reservation in the child arena (full arena -> step stays PENDING and the
// next sweep retries, natural backpressure like max_parallel windowing)
__global__ __launch_bounds__(BLOCK) void wf_expand_kernel(
    const int* __restrict__ disp_runs,
    const int* __restrict__ disp_steps,
    const int* __restrict__ disp_count,
    unsigned char* __restrict__ step_state,   // [NR*64]
    int* __restrict__ children_todo,          // [NR*64] children left to emit
    int* __restrict__ children_out,           // [NR*64] in flight
    const int* __restrict__ max_parallel,     // [NR*64] 0 = unlimited
    int* __restrict__ child_tag,              // [CB] run*64+step
    int* __restrict__ child_seq,              // [CB] per-step emission ordinal
    int* __restrict__ child_widx,             // [CB] global worker pick
    int* __restrict__ child_count,            // [1]
    int* __restrict__ children_emitted,       // [NR*64] monotone ordinal source
    int* __restrict__ dispatch_tick,          // [NR*64] stamped for K4-WF
    const int* __restrict__ tick_p,           // [1]
    const int* __restrict__ order,            // [NWG] spread order (K2c)
    const int* __restrict__ valid_count,      // [1]
    int disp_cap, int CB)
{
    const int e = (blockIdx.x * BLOCK + threadIdx.x) / WAVE;
    const int lane = threadIdx.x % WAVE;
    if (e >= min(*disp_count, disp_cap)) return;
    const int run = disp_runs[e];
    const int step = disp_steps[e];
    const size_t rs = (size_t)run * 64 + step;
    int base = 0, todo = 0, seq0 = 0;
    if (lane == 0) {
        todo = children_todo[rs];
        // for_each max_parallel windowing (dataflow_test.go:71 semantics):
        // emit only up to the window; the commit pass re-opens the step as
        // soon as in-flight drops below the window
        const int maxp = max_parallel[rs];
        if (maxp > 0) {
            const int room = maxp - children_out[rs];
            if (room < todo) todo = room > 0 ? room : 0;
        }
        // monotone per-step emission counter: a deterministic per-child
        // ordinal stream identical on every backend regardless of atomic
        // packing order (retried children get FRESH ordinals)
        seq0 = children_emitted[rs];
        if (todo > 0) {
            base = atomicAdd(child_count, todo);
            if (base + todo > CB) {      // arena full: roll back, retry later
                atomicSub(child_count, todo);
                todo = -1;
            } else {
                step_state[rs] = WFS_DISPATCHED;
                children_out[rs] += todo;
                children_emitted[rs] += todo;
                children_todo[rs] -= todo;
                dispatch_tick[rs] = *tick_p;
            }
        } else if (children_out[rs] > 0) {
            // window full: mark DISPATCHED so the sweep stops re-emitting
            // until the commit pass re-opens it
            step_state[rs] = WFS_DISPATCHED;
        }
    }
    base = __shfl(base, 0, WAVE);
    todo = __shfl(todo, 0, WAVE);
    seq0 = __shfl(seq0, 0, WAVE);
    if (todo <= 0) return;
    const int tag = run * 64 + step;
    const int V = max(1, *valid_count);
    for (int k = lane; k < todo; k += WAVE) {
        child_tag[base + k] = tag;
        child_seq[base + k] = seq0 + k;
        // spread: children are unconstrained -> round-robin the least-loaded
        // order (K2c semantics, deterministic)
        child_widx[base + k] = order[(base + k) % min(V, 1024)];
    }
}

// deterministic per-entry failure injection (config #5 retry waves): a
// splitmix-style hash of (tag, tick, arena position) -> fail if < fail_ppt
// per mille. The CPU oracle mirrors this bit-for-bit.
__device__ __forceinline__ unsigned int wf_mix(unsigned int x) {
    x ^= x >> 16; x *= 0x7feb352du; x ^= x >> 15; x *= 0x846ca68bu; x ^= x >> 16;
    return x;
}

// apply child results over the packed send segments (the owner rank's view
// of what it dispatched this tick; echo workers are deterministic-success,
// failures are injected by a hash of the child's (tag, emission ordinal) —
// independent of packing order, so every backend agrees). Flagged entries
// are redeliveries whose tag/seq live in the rq_prev carries.
__global__ __launch_bounds__(BLOCK) void wf_apply_kernel(
    const int* __restrict__ send_slots,       // [world*cap]
    const int* __restrict__ send_cnt,         // [world]
    const int* __restrict__ child_tag,        // [CB]
    const int* __restrict__ child_seq,        // [CB]
    const int* __restrict__ rq_prev_tag,      // [rq_cap]
    const int* __restrict__ rq_prev_seq,      // [rq_cap]
    int* __restrict__ children_done,          // [NR*64]
    int* __restrict__ children_fail,          // [NR*64]
    int* __restrict__ children_out,           // [NR*64]
    int fail_ppt, int drop_ppt, int cap, int world)
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    if (i >= world * cap) return;
    if ((i % cap) >= min(send_cnt[i / cap], cap)) return;
    const int sslot = send_slots[i];
    const int tag = sslot >= 0 ? child_tag[sslot] : rq_prev_tag[-1 - sslot];
    const int seq = sslot >= 0 ? child_seq[sslot] : rq_prev_seq[-1 - sslot];
    // lost-result injection (crashed worker): the child simply vanishes;
    // children_out stays up and the K4-WF timeout scan recovers it later
    const unsigned int hd = wf_mix((unsigned int)tag * 0x85EBCA6Bu
                                   ^ (unsigned int)seq * 0xC2B2AE35u);
    if ((int)(hd % 1000u) < drop_ppt) return;
    const unsigned int h = wf_mix((unsigned int)tag * 2654435761u
                                  ^ (unsigned int)seq * 40503u);
    if ((int)(h % 1000u) < fail_ppt) atomicAdd(&children_fail[tag], 1);
    else atomicAdd(&children_done[tag], 1);
    atomicSub(&children_out[tag], 1);
}

// dead-letter application: entries dropped by the requeue ring (max-deliver
// exceeded / ring full) count as failed children so their steps can retry or
// fail instead of hanging forever
__global__ __launch_bounds__(BLOCK) void wf_apply_dead_kernel(
    const int* __restrict__ dead_src,         // [dcap] slot or -1-j flags
    const int* __restrict__ dead_count,       // [1]
    const int* __restrict__ child_tag,
    const int* __restrict__ rq_prev_tag,
    int* __restrict__ children_fail,
    int* __restrict__ children_out,
    int dcap)
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    if (i >= min(*dead_count, dcap)) return;
    const int s = dead_src[i];
    const int tag = s >= 0 ? child_tag[s] : rq_prev_tag[-1 - s];
    atomicAdd(&children_fail[tag], 1);
    atomicSub(&children_out[tag], 1);
}

// K4-WF: stale-step timeout scan (reconciler.go:88-144 analog over the
// run/step table). A step DISPATCHED for > cutoff ticks with children still
// outstanding has lost them (worker crash / dropped result): the remainder
// is declared TIMEOUT — counted, converted to failed children — so the
// commit pass retries them with backoff instead of hanging forever.
__global__ __launch_bounds__(BLOCK) void wf_timeout_scan_kernel(
    const unsigned char* __restrict__ step_state,
    int* __restrict__ children_out,
    int* __restrict__ children_fail,
    const int* __restrict__ dispatch_tick,    // [NR*64] last expansion tick
    const int* __restrict__ tick_p, int cutoff,
    unsigned long long* __restrict__ timeout_count,
    int NRS)
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    if (i >= NRS) return;
    if (step_state[i] != WFS_DISPATCHED) return;
    const int lost = children_out[i];
    if (lost <= 0) return;
    if (dispatch_tick[i] > *tick_p - cutoff) return;
    children_fail[i] += lost;
    children_out[i] = 0;
    atomicAdd(timeout_count, (unsigned long long)lost);
}

// continuous re-admission (config #5 "hold 1M concurrent runs"): terminal
// runs in `filter_state` are reset from the creation template and re-enter
// the system. SUCCEEDED runs re-admit on device every tick; FAILED runs
// wait for the host DLQ drain (dlq_store.go analog) which re-admits them
// through the same kernel after recording the entries.
__global__ __launch_bounds__(BLOCK) void wf_readmit_kernel(
    unsigned char* __restrict__ run_active,
    unsigned char* __restrict__ run_state,
    const unsigned char* __restrict__ n_steps,
    unsigned char* __restrict__ step_state,
    int* __restrict__ step_attempts,
    int* __restrict__ children_todo,
    int* __restrict__ children_out,
    int* __restrict__ children_done,
    int* __restrict__ children_fail,
    int* __restrict__ children_emitted,
    int* __restrict__ next_ready,
    int* __restrict__ dispatch_tick,
    const int* __restrict__ todo_tmpl,        // [NR*64]
    const int* __restrict__ nready_tmpl,      // [NR*64]
    int filter_state,
    unsigned long long* __restrict__ admit_count,
    int NR)
{
    const int run = blockIdx.x * BLOCK + threadIdx.x;
    if (run >= NR) return;
    if (run_active[run] || run_state[run] != filter_state) return;
    const int ns = n_steps[run];
    const size_t base = (size_t)run * 64;
    for (int s = 0; s < ns; ++s) {
        const size_t i = base + s;
        step_state[i] = WFS_PENDING;
        step_attempts[i] = 0;
        children_todo[i] = todo_tmpl[i];
        children_out[i] = 0;
        children_done[i] = 0;
        children_fail[i] = 0;
        children_emitted[i] = 0;
        next_ready[i] = nready_tmpl[i];
        dispatch_tick[i] = 0;
    }
    run_state[run] = 0;
    run_active[run] = 1;
    atomicAdd(admit_count, 1ull);
}

// step commit: aggregate children (engine.go:1623-1645) + retry/backoff
// (computeBackoff :1573-1595, attempts capped by max_retries)
__global__ __launch_bounds__(BLOCK) void wf_commit_kernel(
    unsigned char* __restrict__ step_state,   // [NR*64]
    int* __restrict__ step_attempts,          // [NR*64]
    int* __restrict__ children_todo,
    int* __restrict__ children_out,
    int* __restrict__ children_done,          // kept: successes accumulate
    int* __restrict__ children_fail,
    const int* __restrict__ max_parallel,
    int* __restrict__ next_ready,
    const int* __restrict__ tick_p, int max_retries,
    unsigned long long* __restrict__ retry_count,  // cumulative children retried
    int NRS)
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    if (i >= NRS) return;
    if (step_state[i] != WFS_DISPATCHED) return;
    if (children_out[i] != 0) {
        // window slide: with children still in flight, re-open emission as
        // soon as there is BOTH work left and room (max_parallel window, or
        // freed child-arena space for the unlimited case)
        if (children_todo[i] > 0 &&
            (max_parallel[i] == 0 || children_out[i] < max_parallel[i]))
            step_state[i] = WFS_PENDING;
        return;
    }
    const int fail = children_fail[i];
    if (fail > 0) {
        if (step_attempts[i] < max_retries) {
            step_attempts[i] += 1;
            children_todo[i] += fail;      // only failed children re-run
            children_fail[i] = 0;
            next_ready[i] = *tick_p + min(1 << step_attempts[i], 16);
            step_state[i] = WFS_PENDING;
            atomicAdd(retry_count, (unsigned long long)fail);
        } else {
            step_state[i] = WFS_FAILED;
        }
    } else if (children_todo[i] > 0) {
        step_state[i] = WFS_PENDING;       // partial emission resumes
    } else {
        step_state[i] = WFS_SUCCEEDED;
    }
}

// run status roll-up (updateRunStatus :1647-1699): all steps SUCCEEDED or
// SKIPPED -> run SUCCEEDED; any FAILED step -> run FAILED (retries already
// exhausted at step level). counts[0] += succeeded, counts[1] += failed.
__global__ __launch_bounds__(BLOCK) void wf_status_kernel(
    const unsigned char* __restrict__ step_state,
    const unsigned char* __restrict__ n_steps,
    unsigned char* __restrict__ run_active,
    unsigned char* __restrict__ run_state,    // [NR] WFS_* of the run
    unsigned long long* __restrict__ counts,  // [2]
    int NR)
{
    const int run = blockIdx.x * BLOCK + threadIdx.x;
    if (run >= NR || !run_active[run]) return;
    const int ns = n_steps[run];
    const unsigned char* st = &step_state[(size_t)run * 64];
    bool all_ok = true, any_fail = false;
    for (int s = 0; s < ns; ++s) {
        const unsigned char v = st[s];
        any_fail |= (v == WFS_FAILED);
        all_ok &= (v == WFS_SUCCEEDED || v == WFS_SKIPPED);
    }
    if (any_fail) {
        run_active[run] = 0;
        run_state[run] = WFS_FAILED;
        atomicAdd(&counts[1], 1ull);
    } else if (all_ok) {
        run_active[run] = 0;
        run_state[run] = WFS_SUCCEEDED;
        atomicAdd(&counts[0], 1ull);
    }
}

// host-driven approval grants (the admin hop): verdict 1 -> SUCCEEDED,
// 0 -> FAILED (rejection fails the step, which fails the run)
__global__ __launch_bounds__(BLOCK) void wf_grant_kernel(
    const int* __restrict__ grant_runs,
    const int* __restrict__ grant_steps,
    const unsigned char* __restrict__ verdicts,
    int n,
    unsigned char* __restrict__ step_state)
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    if (i >= n) return;
    const size_t rs = (size_t)grant_runs[i] * 64 + grant_steps[i];
    if (step_state[rs] == WFS_WAITING)
        step_state[rs] = verdicts[i] ? WFS_SUCCEEDED : WFS_FAILED;
}

// ---------------------------------------------------------------------------
// K4: deadline / staleness scan -> TIMEOUT candidates (compacted list)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void deadline_scan_kernel(
    const unsigned char* __restrict__ states,   // [N]
    const long long* __restrict__ deadlines,    // [N] unix micros
    const long long* __restrict__ updated_at,   // [N] unix micros
    long long now_us, long long dispatch_cutoff_us, long long running_cutoff_us,
    int* __restrict__ out_slots,                // [cap]
    int* __restrict__ out_count,                // [1]
    int N, int cap)
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    if (i >= N) return;
    const unsigned char st = states[i];
    // active states: PENDING(1) APPROVAL(2) SCHEDULED(3) DISPATCHED(4) RUNNING(5)
    bool expired = false;
    if (st >= 1 && st <= 5 && deadlines[i] <= now_us) expired = true;
    if (st == 4 && updated_at[i] <= dispatch_cutoff_us) expired = true;  // DISPATCHED stale
    if (st == 5 && updated_at[i] <= running_cutoff_us) expired = true;   // RUNNING stale
    const int lane = threadIdx.x % WAVE;
    const int pos = wave_append_slot(expired, out_count, lane);
    if (pos >= 0 && pos < cap) out_slots[pos] = i;
}

// ---------------------------------------------------------------------------
// Echo worker: touch the job payload, write a result word, bump counters
// ---------------------------------------------------------------------------
// Payload arena: flat uint32 arena, job i owns [i*stride, (i+1)*stride).
// The "work" is a full read of the payload (the echo semantics: the result
// is derived from the entire context blob) + one result word per job.
__global__ __launch_bounds__(BLOCK) void echo_worker_kernel(
    const unsigned int* __restrict__ ctx_arena,  // [B*stride]
    unsigned int* __restrict__ res_arena,        // [B*stride]
    unsigned int* __restrict__ res_sum,          // [B]
    int B, int stride)
{
    // one wave per job: lanes stream the payload, wave-reduce a checksum,
    // and echo the payload into the result arena (memory-bound, like the
    // reference's echo worker copying ctx -> res through Redis).
    const int wave_id = (blockIdx.x * BLOCK + threadIdx.x) / WAVE;
    const int lane = threadIdx.x % WAVE;
    if (wave_id >= B) return;
    const size_t basep = (size_t)wave_id * stride;
    unsigned int acc = 0;
    for (int k = lane; k < stride; k += WAVE) {
        const unsigned int v = ctx_arena[basep + k];
        res_arena[basep + k] = v;
        acc += v;
    }
    // wave reduction
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        acc += __shfl_down(acc, off, WAVE);
    if (lane == 0) res_sum[wave_id] = acc;
}

__global__ __launch_bounds__(BLOCK) void echo_worker_indexed_kernel(
    const unsigned int* __restrict__ ctx_arena,  // [N*stride] home arena
    const int* __restrict__ slots,               // [B] job slots to execute
    unsigned int* __restrict__ res_arena,        // [N*stride] result arena
    unsigned int* __restrict__ res_sum,          // [N] by slot
    int B, int stride)
{
    // local-dispatch echo: read the job's payload in place (no staging copy
    // — within one GPU "dispatch" is just ownership, exactly as the worker
    // pool reads the HBM arena directly), write result + checksum.
    const int w = (blockIdx.x * BLOCK + threadIdx.x) / WAVE;
    const int lane = threadIdx.x % WAVE;
    if (w >= B) return;
    const int slot = slots[w];
    const size_t basep = (size_t)slot * stride;
    unsigned int acc = 0;
    for (int k = lane; k < stride; k += WAVE) {
        const unsigned int v = ctx_arena[basep + k];
        res_arena[basep + k] = v;
        acc += v;
    }
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        acc += __shfl_down(acc, off, WAVE);
    if (lane == 0) res_sum[slot] = acc;
}


// ---------------------------------------------------------------------------
// K1-MFMA: policy first-match on the matrix cores (comparison variant)
// ---------------------------------------------------------------------------
// One v_mfma_i32_16x16x64_i8 per dimension per 16-job x 16-rule tile computes
// all 256 set-intersection cardinalities of that dim at once. Epilogue turns
// the 9 dot products into the match predicate and reduces first-match per
// job across the 16 rule columns with 16-lane shuffles.
// See ops/policy_mfma.py for the fragment prepack and the measured
// bitset-vs-MFMA verdict.
typedef int v4i __attribute__((ext_vector_type(4)));

__global__ __launch_bounds__(BLOCK) void policy_first_match_mfma_kernel(
    const signed char* __restrict__ a_pack,   // [Jt][9][64][16]
    const signed char* __restrict__ b_pack,   // [Rt][9][64][16]
    const int* __restrict__ cards,            // [Rt*16][9]
    const signed char* __restrict__ rule_secrets, // [Rt*16]
    const unsigned char* __restrict__ job_secrets, // [J]
    const int* __restrict__ tile_dims,        // [Rt] bitmask of live dims
    int* __restrict__ out_first,              // [J] pre-filled INT_MAX
    int J, int R, int tiles_per_chunk)
{
    const int wave = threadIdx.x / WAVE;      // 4 waves per block
    const int lane = threadIdx.x % WAVE;
    const int jt = blockIdx.x;                // one job tile per block
    const int Rt = (R + 15) / 16;
    const int chunk_begin = blockIdx.y * tiles_per_chunk;
    const int chunk_end = min(Rt, chunk_begin + tiles_per_chunk);

    // A fragments for the 9 dims: lane-linear 16 B loads
    v4i afrag[9];
    #pragma unroll
    for (int d = 0; d < 9; ++d)
        afrag[d] = *(const v4i*)&a_pack[(((size_t)jt * 9 + d) * 64 + lane) * 16];

    const int row_base = (lane >> 4) * 4;     // 4 job rows per lane
    const int col = lane & 15;                // rule column
    int best[4] = {INT_MAX, INT_MAX, INT_MAX, INT_MAX};
    unsigned char jsec[4];
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
        const int job = jt * 16 + row_base + r;
        jsec[r] = (job < J) ? job_secrets[job] : 0;
    }

    // early-exit checks only pay when each wave scans enough rule tiles
    const bool do_exit = (chunk_end - chunk_begin) > 32;

    for (int rt = chunk_begin + wave; rt < chunk_end; rt += 4) {
        // dim skipping: most rules constrain only 2-4 of the 9 set dims; a
        // dim whose card is 0 for EVERY rule in the tile auto-passes (any-of:
        // card==0 | _; all-of: 0==0), so its MFMA + B-fragment load is dead.
        // tile_dims is the per-tile union, uniform across the wave.
        const unsigned dmask = (unsigned)tile_dims[rt];
        const int rule = rt * 16 + col;
        const signed char rsec = rule_secrets[rule];
        bool ok[4] = {true, true, true, true};
        #pragma unroll
        for (int d = 0; d < 9; ++d) {
            if (!(dmask & (1u << d))) continue;
            const v4i bfrag = *(const v4i*)&b_pack[(((size_t)rt * 9 + d) * 64 + lane) * 16];
            v4i zero = {0, 0, 0, 0};
            const v4i acc = __builtin_amdgcn_mfma_i32_16x16x64_i8(afrag[d], bfrag, zero, 0, 0, 0);
            const int card = cards[rule * 9 + d];
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                if (d < 7) ok[r] &= (card == 0) | (acc[r] > 0);   // any-of
                else       ok[r] &= (acc[r] == card);             // all-of
            }
        }
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            const bool pass = ok[r] & ((rsec < 0) | (rsec == (signed char)jsec[r]));
            if (pass && rule < R) best[r] = min(best[r], rule);
        }

        // first-match early exit, adaptive: stop once every job row of the
        // tile has SOME match — later tiles only yield larger rule ids, and
        // a smaller id found by another wave/block wins at the atomicMin
        // anyway. Folding out_first (other blocks' published finds; the
        // value only decreases, so a stale read just skips less) pays only
        // when each wave scans many tiles: measured on the default config
        // (8 tiles/wave) ANY per-tile check loses (171.6 -> 145-167M
        // jobs/s), while at 16k rules (128 tiles/wave) the fold wins 35%+.
        // So: no checks on short scans, fold+ballot every 4th tile on long.
        if (do_exit && ((rt >> 2) & 3) == 3) {
            bool all_matched = true;
            #pragma unroll
            for (int r = 0; r < 4; ++r) {
                const int job = jt * 16 + row_base + r;
                if (job < J)
                    best[r] = min(best[r], __hip_atomic_load(&out_first[job],
                        __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT));
                const unsigned long long m = __ballot(best[r] != INT_MAX);
                all_matched &= ((m & 0xFFFFull) != 0) & (((m >> 16) & 0xFFFFull) != 0)
                             & (((m >> 32) & 0xFFFFull) != 0) & (((m >> 48) & 0xFFFFull) != 0);
            }
            if (all_matched) break;
        }
    }

    // min across the 16 rule columns (lanes sharing the same lane>>4 group)
    #pragma unroll
    for (int off = 8; off > 0; off >>= 1) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int other = __shfl_xor(best[r], off, WAVE);
            if ((lane & 15) < 16) best[r] = min(best[r], other);
        }
    }
    if ((lane & 15) == 0) {
        #pragma unroll
        for (int r = 0; r < 4; ++r) {
            const int job = jt * 16 + row_base + r;
            if (job < J && best[r] != INT_MAX) atomicMin(&out_first[job], best[r]);
        }
    }
}

// ---------------------------------------------------------------------------
// Tick-fusion kernels: device-side compaction + count-pointer variants so a
// whole single-GPU control-plane tick is a fixed kernel sequence (no host
// syncs) and can be captured into a hipGraph (guide §Guideline 9).
// ---------------------------------------------------------------------------

// decisions gather + allow/deny split (replaces 6 torch glue ops)
__global__ __launch_bounds__(BLOCK) void policy_gate_kernel(
    const int* __restrict__ first,            // [J] first-match rule (-1 none)
    const signed char* __restrict__ decisions, // [R]
    signed char* __restrict__ out_decision,   // [J]
    int* __restrict__ denied_slots,           // [J] compacted
    int* __restrict__ denied_count,           // [1]
    int* __restrict__ allowed_slots,          // [J] compacted
    int* __restrict__ allowed_count,          // [1]
    int J)
{
    const int j = blockIdx.x * BLOCK + threadIdx.x;
    const int lane = threadIdx.x % WAVE;
    const bool live = j < J;
    signed char d = 0;
    if (live) {
        const int r = first[j];
        d = (r >= 0) ? decisions[r] : (signed char)1;  // default allow
        out_decision[j] = d;
    }
    const bool allowed = live && (d == 1 || d == 5);
    const int apos = wave_append_slot(allowed, allowed_count, lane);
    if (apos >= 0) allowed_slots[apos] = j;
    const bool denied = live && !allowed;
    const int dpos = wave_append_slot(denied, denied_count, lane);
    if (dpos >= 0) denied_slots[dpos] = j;
}

// gate + DENIED transition + DLQ ring in ONE launch: the profile showed the
// split (gate, apply_transitions_dyn[DENIED], dlq_ring_append) spending
// ~14 us/tick on three ~4.6 us launches over the same denied set
// (profiles/r14_bench_ktrace_stats.txt) — at a 150 us tick that is pure
// launch-boundary tax (MI355X_MICROARCH.md price-list "kernel launch").
__global__ __launch_bounds__(BLOCK) void policy_gate_full_kernel(
    const int* __restrict__ first,            // [J]
    const signed char* __restrict__ decisions, // [R]
    signed char* __restrict__ out_decision,   // [J]
    int* __restrict__ denied_slots,           // [J] compacted
    int* __restrict__ denied_count,           // [1]
    int* __restrict__ allowed_slots,          // [J] compacted
    int* __restrict__ allowed_count,          // [1]
    unsigned char* __restrict__ states,
    long long* __restrict__ deadlines,
    int* __restrict__ dlq_ring,               // [ring_size]
    int* __restrict__ dlq_head,               // [1] monotonic
    int ring_size,
    int J, int R)
{
    const int j = blockIdx.x * BLOCK + threadIdx.x;
    const int lane = threadIdx.x % WAVE;
    const bool live = j < J;
    signed char d = 0;
    if (live) {
        const int r = first[j];
        // no-match is -1 (host-masked) OR the raw INT_MAX atomicMin identity
        // when the producer skipped the mask pass (_into variants)
        d = (r >= 0 && r < R) ? decisions[r] : (signed char)1;  // default allow
        out_decision[j] = d;
    }
    const bool allowed = live && (d == 1 || d == 5);
    const int apos = wave_append_slot(allowed, allowed_count, lane);
    if (apos >= 0) allowed_slots[apos] = j;
    const bool denied = live && !allowed;
    const int dpos = wave_append_slot(denied, denied_count, lane);
    if (dpos >= 0) denied_slots[dpos] = j;
    // DENIED transition (LUT-checked) + terminal deadline clear, fused
    bool dlq = false;
    if (denied) {
        const unsigned char from = states[j];
        if (d_transition_lut[from * N_STATES + 10]) {  // -> DENIED
            states[j] = 10;
            deadlines[j] = (long long)0x7fffffffffffffffLL;
            dlq = true;
        }
    }
    const int rpos = wave_append_slot(dlq, dlq_head, lane);
    if (rpos >= 0) dlq_ring[rpos % ring_size] = j;
}

// one-launch per-tick reset: states + the small counter words (replaces two
// torch zero_() elementwise launches inside the captured tick)
__global__ __launch_bounds__(BLOCK) void tick_reset_kernel(
    unsigned char* __restrict__ states, int B,
    int* __restrict__ counts, int ncounts)
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    if (i < B) states[i] = 0;
    if (i < ncounts) counts[i] = 0;
}

// tick prologue in one launch: every ring slot is re-admitted as PENDING
// (CREATED -> PENDING is the only LUT edge exercised by the old zero_() +
// apply_transitions pair, so write it directly) and the counter words reset
__global__ __launch_bounds__(BLOCK) void begin_tick_kernel(
    unsigned char* __restrict__ states, int B,
    int* __restrict__ counts, int ncounts,
    int* __restrict__ first)               // [B] primed for atomicMin, or null
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    if (i < B) {
        states[i] = 1;  // PENDING
        if (first != nullptr) first[i] = INT_MAX;
    }
    if (i < ncounts) counts[i] = 0;
}

// end-of-tick stats fold: the per-tick D2H counts read was the tick's only
// host sync (~15 us graph-replay + copy stall); accumulate on-device and let
// the host read ONCE per timed window instead
__global__ void accumulate_counts_kernel(
    const int* __restrict__ counts, long long* __restrict__ acc, int n)
{
    const int t = threadIdx.x;
    if (t < n) acc[t] += (long long)counts[t];
}

// routable compaction with the K2c spread computed inline: spread_pick keyed
// the round-robin on the slot index j, which this kernel already has — so a
// separate spread launch over all B slots was redundant work + launch tax
__global__ __launch_bounds__(BLOCK) void compact_routable_spread_kernel(
    const int* __restrict__ allowed_slots,    // [<=J]
    const int* __restrict__ allowed_count,    // [1]
    const int* __restrict__ pick,             // [J] exact least-loaded pick
    const int* __restrict__ order,            // [NW] workers by key asc
    const int* __restrict__ valid_count,      // [1] non-overloaded workers
    const long long* __restrict__ j_poolmask,
    const long long* __restrict__ j_labels,
    long long full_mask,
    int K,
    int* __restrict__ routable_slots,         // out
    int* __restrict__ routable_widx,          // out (global worker idx)
    int* __restrict__ routable_count)         // [1]
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    const int lane = threadIdx.x % WAVE;
    const bool live = i < *allowed_count;
    int j = 0, w = -1;
    if (live) {
        j = allowed_slots[i];
        if (j_labels[j] == 0 && j_poolmask[j] == full_mask) {
            const int V = min(*valid_count, K);
            w = (V > 0) ? order[j % V] : pick[j];
        } else {
            w = pick[j];  // constrained: keep the exact scan pick
        }
    }
    const int pos = wave_append_slot(live && w >= 0, routable_count, lane);
    if (pos >= 0) {
        routable_slots[pos] = j;
        routable_widx[pos] = w;
    }
}

// routable compaction: allowed jobs with a worker pick
__global__ __launch_bounds__(BLOCK) void compact_routable_kernel(
    const int* __restrict__ allowed_slots,    // [<=J]
    const int* __restrict__ allowed_count,    // [1]
    const int* __restrict__ pick,             // [J]
    int* __restrict__ routable_slots,         // out
    int* __restrict__ routable_widx,          // out (global worker idx)
    int* __restrict__ routable_count)         // [1]
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    const int lane = threadIdx.x % WAVE;
    const bool live = i < *allowed_count;
    int j = 0, w = -1;
    if (live) {
        j = allowed_slots[i];
        w = pick[j];
    }
    const int pos = wave_append_slot(live && w >= 0, routable_count, lane);
    if (pos >= 0) {
        routable_slots[pos] = j;
        routable_widx[pos] = w;
    }
}

// K5 with device-resident count + uniform target state
__global__ __launch_bounds__(BLOCK) void apply_transitions_dyn_kernel(
    unsigned char* __restrict__ states,
    int* __restrict__ attempts,
    long long* __restrict__ deadlines,
    const int* __restrict__ slots,
    const int* __restrict__ count,            // [1]
    unsigned char to)
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    if (i >= *count) return;
    const int slot = slots[i];
    const unsigned char from = states[slot];
    if (!d_transition_lut[from * N_STATES + to]) return;
    states[slot] = to;
    if (to == 3 && from != 3) attempts[slot] += 1;
    if (to >= 6) deadlines[slot] = (long long)0x7fffffffffffffffLL;
}

// chained K5: march each routed slot through up to 4 states in ONE launch
// (SCHEDULED -> DISPATCHED -> RUNNING -> SUCCEEDED after the echo worker; the
// intermediate states are unobservable inside a captured graph, so four
// launches over the identical slot set were pure launch tax — see
// profiles/r14_bench_ktrace_stats.txt: apply_transitions_dyn 5x/tick, 13.6%).
// Also zeroes an extra int region (the per-tick local-load accumulator) so
// the load_feedback pass starts clean without its own zero_() launch.
__global__ __launch_bounds__(BLOCK) void apply_transitions_chain_dyn_kernel(
    unsigned char* __restrict__ states,
    int* __restrict__ attempts,
    long long* __restrict__ deadlines,
    const int* __restrict__ slots,
    const int* __restrict__ count,            // [1]
    int to0, int to1, int to2, int to3,       // -1 = unused
    int* __restrict__ extra_zero, int n_extra)
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    if (i < n_extra) extra_zero[i] = 0;
    if (i >= *count) return;
    const int slot = slots[i];
    unsigned char from = states[slot];
    const int chain[4] = {to0, to1, to2, to3};
    #pragma unroll
    for (int c = 0; c < 4; ++c) {
        if (chain[c] < 0) break;
        const unsigned char to = (unsigned char)chain[c];
        if (!d_transition_lut[from * N_STATES + to]) break;
        if (to == 3 && from != 3) attempts[slot] += 1;
        from = to;
    }
    states[slot] = from;
    if (from >= 6) deadlines[slot] = (long long)0x7fffffffffffffffLL;
}

__global__ __launch_bounds__(BLOCK) void echo_worker_indexed_dyn_kernel(
    const unsigned int* __restrict__ ctx_arena,
    const int* __restrict__ slots,
    const int* __restrict__ count,            // [1]
    unsigned int* __restrict__ res_arena,
    unsigned int* __restrict__ res_sum,
    int stride)
{
    const int w = (blockIdx.x * BLOCK + threadIdx.x) / WAVE;
    const int lane = threadIdx.x % WAVE;
    if (w >= *count) return;
    const int slot = slots[w];
    const size_t basep = (size_t)slot * stride;
    unsigned int acc = 0;
    for (int k = lane; k < stride; k += WAVE) {
        const unsigned int v = ctx_arena[basep + k];
        res_arena[basep + k] = v;
        acc += v;
    }
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        acc += __shfl_down(acc, off, WAVE);
    if (lane == 0) res_sum[slot] = acc;
}

// per-worker active-count histogram from the routable list
__global__ __launch_bounds__(BLOCK) void load_feedback_kernel(
    const int* __restrict__ routable_widx,
    const int* __restrict__ count,
    int* __restrict__ w_active_local,         // [NWL] pre-zeroed
    int nwl, int my_rank)
{
    // wave match-any aggregation: a frozen-snapshot least-loaded pick is
    // highly concentrated (often ONE worker for a whole homogeneous batch),
    // so per-lane atomics serialize on a single word (~88 atomics/us).
    // Group lanes by equal widx and issue one atomicAdd per distinct value.
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    const int lane = threadIdx.x % WAVE;
    bool live = i < *count;
    int bin = -1;
    if (live) {
        const int w = routable_widx[i];
        if (w / nwl == my_rank) bin = w % nwl;
        else live = false;
    }
    unsigned long long pending = __ballot(live);
    while (pending) {
        const int leader = __ffsll((long long)pending) - 1;
        const int leader_bin = __shfl(bin, leader, WAVE);
        const unsigned long long same = __ballot(live && bin == leader_bin);
        if (lane == leader) atomicAdd(&w_active_local[leader_bin], __popcll(same));
        pending &= ~same;
    }
}


// ---------------------------------------------------------------------------
// Padded cross-rank dispatch (multi-GPU path): fixed-capacity per-destination
// segments so the all_to_all shapes are static and the tick needs NO host
// splits sync. Capacity per destination = B (worst case: the whole batch to
// one rank); validity travels as a device-resident count vector.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(BLOCK) void pack_by_dest_kernel(
    const int* __restrict__ routable_slots,   // [B]
    const int* __restrict__ routable_widx,    // [B] global worker idx
    const int* __restrict__ routable_count,   // [1]
    int* __restrict__ send_slots,             // [world*B]
    int* __restrict__ send_widx,              // [world*B] local worker idx
    int* __restrict__ send_cnt,               // [world] pre-zeroed
    int nwl, int cap,
    int* __restrict__ rq_src,                 // [rq_cap] requeue ring: slot (this batch)
    int* __restrict__ rq_widx,                // [rq_cap] global widx
    int* __restrict__ rq_attempts,            // [rq_cap] delivery attempts
    int* __restrict__ rq_count,               // [1]
    unsigned long long* __restrict__ rq_dead, // [1] dropped after max attempts / ring full
    int rq_cap,
    int* __restrict__ dead_src,               // [dead_cap] per-tick dead list (DLQ-bound)
    int* __restrict__ dead_count,             // [1] pre-zeroed per tick
    int dead_cap)
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    const int lane = threadIdx.x % WAVE;
    bool live = i < *routable_count;
    int slot = 0, widx = 0, dest = -1;
    if (live) {
        slot = routable_slots[i];
        widx = routable_widx[i];
        dest = widx / nwl;
    }
    // match-any aggregation on dest: one atomicAdd per distinct dest per wave
    unsigned long long pending = __ballot(live);
    int my_pos = -1;
    while (pending) {
        const int leader = __ffsll((long long)pending) - 1;
        const int leader_dest = __shfl(dest, leader, WAVE);
        const unsigned long long same = __ballot(live && dest == leader_dest);
        int base = 0;
        if (lane == leader) base = atomicAdd(&send_cnt[leader_dest], __popcll(same));
        base = __shfl(base, leader, WAVE);
        if (live && dest == leader_dest) {
            my_pos = base + __popcll(same & ((1ull << lane) - 1ull));
            live = false;
        }
        pending &= ~same;
    }
    if (my_pos < 0) return;
    if (my_pos < cap) {
        send_slots[(size_t)dest * cap + my_pos] = slot;
        send_widx[(size_t)dest * cap + my_pos] = widx % nwl;
    } else {
        // destination segment full: NAK-with-redelivery (bus/nats.go:146-168
        // analog) — park in the requeue ring, re-admitted next tick
        const int p = atomicAdd(rq_count, 1);
        if (p < rq_cap) {
            rq_src[p] = slot;      // payload row in this batch's home arena
            rq_widx[p] = widx;
            rq_attempts[p] = 1;
        } else {
            atomicAdd(rq_dead, 1ull);  // ring full -> DLQ-bound, counted
            const int dp = atomicAdd(dead_count, 1);
            if (dp < dead_cap) dead_src[dp] = slot;
        }
    }
}

// Redelivery pack: drain LAST tick's requeue ring into the padded send
// segments ahead of the fresh batch (redelivered jobs have priority, like
// a NAK'd message re-surfacing before new publishes). Entries that find
// their destination full again go back to the (new) requeue ring with
// attempts+1; past RQ_MAX_DELIVER they are dropped to rq_dead (the DLQ
// analog of JetStream max-deliver).
#define RQ_MAX_DELIVER 5
__global__ __launch_bounds__(BLOCK) void pack_requeue_kernel(
    const int* __restrict__ rq_prev_widx,     // [rq_cap]
    const int* __restrict__ rq_prev_attempts, // [rq_cap]
    const int* __restrict__ rq_prev_count,    // [1]
    int* __restrict__ send_slots,             // [world*cap]
    int* __restrict__ send_widx,              // [world*cap]
    int* __restrict__ send_cnt,               // [world] pre-zeroed
    int nwl, int cap,
    int* __restrict__ rq_src,
    int* __restrict__ rq_widx,
    int* __restrict__ rq_attempts,
    int* __restrict__ rq_count,
    unsigned long long* __restrict__ rq_dead,
    int rq_cap,
    int* __restrict__ dead_src,               // [dead_cap]
    int* __restrict__ dead_count,             // [1] pre-zeroed per tick
    int dead_cap)
{
    const int j = blockIdx.x * BLOCK + threadIdx.x;
    const int n = min(*rq_prev_count, rq_cap);
    if (j >= n) return;
    const int widx = rq_prev_widx[j];
    const int dest = widx / nwl;
    const int pos = atomicAdd(&send_cnt[dest], 1);
    if (pos < cap) {
        // flagged slot: payload comes from rq_prev_payload row j
        send_slots[(size_t)dest * cap + pos] = -1 - j;
        send_widx[(size_t)dest * cap + pos] = widx % nwl;
    } else {
        const int att = rq_prev_attempts[j] + 1;
        bool drop = att > RQ_MAX_DELIVER;
        if (!drop) {
            const int p = atomicAdd(rq_count, 1);
            if (p < rq_cap) {
                rq_src[p] = -1 - j;   // payload row in rq_prev_payload
                rq_widx[p] = widx;
                rq_attempts[p] = att;
            } else {
                drop = true;
            }
        }
        if (drop) {
            atomicAdd(rq_dead, 1ull);
            const int dp = atomicAdd(dead_count, 1);
            if (dp < dead_cap) dead_src[dp] = -1 - j;
        }
    }
}

// Materialize this tick's requeue-ring payloads into their own arena so the
// entries survive home-arena reuse (ring slots rotate every tick): source is
// the current batch (src >= 0) or last tick's rq arena (flagged).
__global__ __launch_bounds__(BLOCK) void materialize_rq_payload_kernel(
    const unsigned int* __restrict__ payload,         // [B*stride] this batch
    const unsigned int* __restrict__ rq_prev_payload, // [rq_cap*stride]
    const int* __restrict__ rq_src,                   // [rq_cap]
    const int* __restrict__ rq_count,                 // [1]
    unsigned int* __restrict__ rq_payload,            // [rq_cap*stride]
    int stride, int rq_cap)
{
    const int j = (blockIdx.x * BLOCK + threadIdx.x) / WAVE;
    const int lane = threadIdx.x % WAVE;
    const int n = min(*rq_count, rq_cap);
    if (j >= n) return;
    const int src = rq_src[j];
    const unsigned int* row = src >= 0 ? payload + (size_t)src * stride
                                       : rq_prev_payload + (size_t)(-1 - src) * stride;
    unsigned int* dst = rq_payload + (size_t)j * stride;
    for (int k = lane; k < stride; k += WAVE)
        dst[k] = row[k];
}

// gather payload rows into the padded send arena, region-valid; flagged
// (negative) slots are redeliveries whose payload lives in rq_prev_payload
__global__ __launch_bounds__(BLOCK) void gather_payload_padded_kernel(
    const unsigned int* __restrict__ payload,  // [B*stride] home arena
    const unsigned int* __restrict__ rq_prev_payload, // [rq_cap*stride]
    const int* __restrict__ send_slots,        // [world*cap]
    const int* __restrict__ send_cnt,          // [world]
    unsigned int* __restrict__ send_payload,   // [world*cap*stride]
    int stride, int cap, int world)
{
    const int w = (blockIdx.x * BLOCK + threadIdx.x) / WAVE;
    const int lane = threadIdx.x % WAVE;
    if (w >= world * cap) return;
    const int region = w / cap, e = w % cap;
    if (e >= send_cnt[region]) return;
    const int sslot = send_slots[w];
    const unsigned int* src = sslot >= 0
        ? payload + (size_t)sslot * stride
        : rq_prev_payload + (size_t)(-1 - sslot) * stride;
    const size_t dst = (size_t)w * stride;
    for (int k = lane; k < stride; k += WAVE)
        send_payload[dst + k] = src[k];
}

// region-valid echo over the padded receive arena
__global__ __launch_bounds__(BLOCK) void echo_padded_kernel(
    const unsigned int* __restrict__ recv_payload, // [world*cap*stride]
    const int* __restrict__ recv_cnt,              // [world]
    unsigned int* __restrict__ res_arena,          // [world*cap*stride]
    unsigned int* __restrict__ res_sums,           // [world*cap]
    int stride, int cap, int world)
{
    const int w = (blockIdx.x * BLOCK + threadIdx.x) / WAVE;
    const int lane = threadIdx.x % WAVE;
    if (w >= world * cap) return;
    const int region = w / cap, e = w % cap;
    if (e >= recv_cnt[region]) return;
    const size_t basep = (size_t)w * stride;
    unsigned int acc = 0;
    for (int k = lane; k < stride; k += WAVE) {
        const unsigned int v = recv_payload[basep + k];
        res_arena[basep + k] = v;
        acc += v;
    }
    #pragma unroll
    for (int off = 32; off > 0; off >>= 1)
        acc += __shfl_down(acc, off, WAVE);
    if (lane == 0) res_sums[w] = acc;
}

// region-valid transitions over the padded slot list
__global__ __launch_bounds__(BLOCK) void apply_transitions_padded_kernel(
    unsigned char* __restrict__ states,
    int* __restrict__ attempts,
    long long* __restrict__ deadlines,
    const int* __restrict__ slots_pad,        // [world*cap]
    const int* __restrict__ cnt,              // [world]
    unsigned char to, int cap, int world)
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    if (i >= world * cap) return;
    if ((i % cap) >= cnt[i / cap]) return;
    const int slot = slots_pad[i];
    if (slot < 0) return;  // redelivered entry: its batch slot was recycled
    const unsigned char from = states[slot];
    if (!d_transition_lut[from * N_STATES + to]) return;
    states[slot] = to;
    if (to == 3 && from != 3) attempts[slot] += 1;
    if (to >= 6) deadlines[slot] = (long long)0x7fffffffffffffffLL;
}

// region-valid per-worker load histogram from padded recv widx
__global__ __launch_bounds__(BLOCK) void load_feedback_padded_kernel(
    const int* __restrict__ recv_widx,        // [world*cap] local worker idx
    const int* __restrict__ recv_cnt,         // [world]
    int* __restrict__ w_active_local,         // [NWL] pre-zeroed
    int cap, int world)
{
    const int i = blockIdx.x * BLOCK + threadIdx.x;
    const int lane = threadIdx.x % WAVE;
    bool live = (i < world * cap) && ((i % cap) < recv_cnt[i / cap]);
    int bin = live ? recv_widx[i] : -1;
    unsigned long long pending = __ballot(live);
    while (pending) {
        const int leader = __ffsll((long long)pending) - 1;
        const int leader_bin = __shfl(bin, leader, WAVE);
        const unsigned long long same = __ballot(live && bin == leader_bin);
        if (lane == leader) atomicAdd(&w_active_local[leader_bin], __popcll(same));
        pending &= ~same;
        live = live && bin != leader_bin;
    }
}

// ---------------------------------------------------------------------------
// Torch extension host wrappers
// ---------------------------------------------------------------------------

// device-side job packing for the MFMA K1 variant: bit words -> A-fragment
// layout (policy_mfma.py _pack_a semantics, byte-for-byte). One thread per
// (job-tile, dim, lane); each expands one 16-bit slice of the job's bit
// word into 16 int8 lanes. Lets the e2e ingest window run the MFMA kernel
// on freshly staged batches without a host packing pass.
__global__ __launch_bounds__(BLOCK) void pack_jobs_mfma_kernel(
    const long long* __restrict__ any_bits,   // [B,7] (1-word vocab)
    const long long* __restrict__ all_bits,   // [B,2]
    signed char* __restrict__ a_pack,         // [Jt,9,64,16]
    int B, int Jt)
{
    const int idx = blockIdx.x * BLOCK + threadIdx.x;
    if (idx >= Jt * 9 * 64) return;
    const int jt = idx / (9 * 64);
    const int rem = idx % (9 * 64);
    const int d = rem / 64;
    const int lane = rem % 64;
    const int row = lane & 15;
    const int kb = lane >> 4;
    const int job = jt * 16 + row;
    unsigned long long w = 0;
    if (job < B)
        w = (unsigned long long)(d < 7 ? any_bits[(size_t)job * 7 + d]
                                       : all_bits[(size_t)job * 2 + (d - 7)]);
    const unsigned int slice = (unsigned int)((w >> (kb * 16)) & 0xFFFFu);
    signed char* out = a_pack + (size_t)idx * 16;
    #pragma unroll
    for (int t = 0; t < 16; ++t)
        out[t] = (signed char)((slice >> t) & 1u);
}

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#define CHECK_DEV(t) TORCH_CHECK((t).is_cuda(), #t " must be on device")
#define CHECK_CONTIG(t) TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

static inline hipStream_t cur_stream() {
    return at::hip::getCurrentHIPStream().stream();
}

torch::Tensor policy_first_match(
    torch::Tensor rule_any, torch::Tensor rule_all, torch::Tensor rule_secrets,
    torch::Tensor rule_mcp, torch::Tensor rule_mcp_any,
    torch::Tensor job_any, torch::Tensor job_all, torch::Tensor job_secrets,
    torch::Tensor job_mcp, torch::Tensor job_mcp_used,
    int64_t rule_chunks)
{
    for (auto* t : {&rule_any, &rule_all, &rule_mcp, &job_any, &job_all, &job_mcp}) {
        CHECK_DEV(*t); CHECK_CONTIG(*t);
    }
    const int W = (int)rule_any.size(2);
    const int R = (int)rule_any.size(0);
    const int J = (int)job_any.size(0);
    TORCH_CHECK(job_any.size(2) == W, "word-count mismatch");
    auto out = torch::full({J}, INT_MAX,
        torch::TensorOptions().dtype(torch::kInt32).device(job_any.device()));
    if (R == 0 || J == 0) {
        return out.masked_fill(out == INT_MAX, -1);
    }
    // JPT: 2 jobs/thread halves the rule-stream traffic per block, but
    // measured NET-NEGATIVE at every shape tried (16k rules: 162->260 us;
    // 100k: 244->323 us — the doubled job-row registers cost occupancy and
    // the per-chunk early-out needs BOTH jobs finished). Kept selectable for
    // future shapes; disabled by measurement.
    const int JPT = 1;
    const int jobs_per_block = BLOCK * JPT;
    const int job_blocks = (J + jobs_per_block - 1) / jobs_per_block;
    int nchunks = (int)rule_chunks;
    if (nchunks <= 0) {
        nchunks = std::max(1, std::min((R + 511) / 512,
                                       std::max(1, 2048 / std::max(job_blocks, 1))));
    }
    const int per_chunk = (R + nchunks - 1) / nchunks;
    hipStream_t stream = cur_stream();

    auto launch = [&](auto kern) {
        hipLaunchKernelGGL(kern, dim3(job_blocks, nchunks), dim3(BLOCK), 0, stream,
            (const long long*)rule_any.data_ptr<int64_t>(),
            (const long long*)rule_all.data_ptr<int64_t>(),
            (const signed char*)rule_secrets.data_ptr<int8_t>(),
            (const long long*)rule_mcp.data_ptr<int64_t>(),
            rule_mcp_any.data_ptr<uint8_t>(),
            (const long long*)job_any.data_ptr<int64_t>(),
            (const long long*)job_all.data_ptr<int64_t>(),
            job_secrets.data_ptr<uint8_t>(),
            (const long long*)job_mcp.data_ptr<int64_t>(),
            job_mcp_used.data_ptr<uint8_t>(),
            out.data_ptr<int>(), R, J, per_chunk);
    };
    if (JPT == 2) {
        switch (W) {
            case 1: launch(policy_first_match_kernel<1, 2>); break;
            case 2: launch(policy_first_match_kernel<2, 2>); break;
            case 4: launch(policy_first_match_kernel<4, 2>); break;
            default: TORCH_CHECK(false, "unsupported word count ", W);
        }
    } else {
        switch (W) {
            case 1: launch(policy_first_match_kernel<1, 1>); break;
            case 2: launch(policy_first_match_kernel<2, 1>); break;
            case 4: launch(policy_first_match_kernel<4, 1>); break;
            default: TORCH_CHECK(false, "unsupported word count ", W);
        }
    }
    out.masked_fill_(out == INT_MAX, -1);
    return out;
}

torch::Tensor worker_precompute(
    torch::Tensor w_pool, torch::Tensor w_active, torch::Tensor w_maxp,
    torch::Tensor w_cpu, torch::Tensor w_gpu)
{
    CHECK_DEV(w_pool);
    const int NW = (int)w_pool.size(0);
    auto keys = torch::empty({NW}, torch::TensorOptions().dtype(torch::kInt64).device(w_pool.device()));
    if (NW == 0) return keys;
    const int blocks = (NW + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(worker_precompute_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        w_pool.data_ptr<int>(), w_active.data_ptr<int>(), w_maxp.data_ptr<int>(),
        w_cpu.data_ptr<float>(), w_gpu.data_ptr<float>(),
        (unsigned long long*)keys.data_ptr<int64_t>(), NW);
    return keys;
}

torch::Tensor least_loaded_pick(
    torch::Tensor w_pool, torch::Tensor w_keys, torch::Tensor w_labels,
    torch::Tensor j_poolmask, torch::Tensor j_labels)
{
    CHECK_DEV(w_pool); CHECK_DEV(j_poolmask);
    const int NW = (int)w_pool.size(0);
    const int NJ = (int)j_poolmask.size(0);
    auto out = torch::empty({NJ}, torch::TensorOptions().dtype(torch::kInt32).device(j_poolmask.device()));
    if (NJ == 0) return out;
    const int jobs_per_block = BLOCK / WAVE;
    const int blocks = (NJ + jobs_per_block - 1) / jobs_per_block;
    hipLaunchKernelGGL(least_loaded_pick_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        w_pool.data_ptr<int>(),
        (const unsigned long long*)w_keys.data_ptr<int64_t>(),
        (const long long*)w_labels.data_ptr<int64_t>(),
        (const long long*)j_poolmask.data_ptr<int64_t>(),
        (const long long*)j_labels.data_ptr<int64_t>(),
        out.data_ptr<int>(), NW, NJ, (const int*)nullptr, 0LL);
    return out;
}
void least_loaded_pick_into(
    torch::Tensor w_pool, torch::Tensor w_keys, torch::Tensor w_labels,
    torch::Tensor j_poolmask, torch::Tensor j_labels,
    torch::Tensor valid_count, int64_t full_mask, torch::Tensor out)
{
    CHECK_DEV(w_pool); CHECK_DEV(j_poolmask);
    const int NW = (int)w_pool.size(0);
    const int NJ = (int)j_poolmask.size(0);
    if (NJ == 0) return;
    const int jobs_per_block = BLOCK / WAVE;
    const int blocks = (NJ + jobs_per_block - 1) / jobs_per_block;
    hipLaunchKernelGGL(least_loaded_pick_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        w_pool.data_ptr<int>(),
        (const unsigned long long*)w_keys.data_ptr<int64_t>(),
        (const long long*)w_labels.data_ptr<int64_t>(),
        (const long long*)j_poolmask.data_ptr<int64_t>(),
        (const long long*)j_labels.data_ptr<int64_t>(),
        out.data_ptr<int>(), NW, NJ,
        valid_count.data_ptr<int>(), (long long)full_mask);
}

void dlq_ring_append(torch::Tensor slots, torch::Tensor count,
                     torch::Tensor ring, torch::Tensor head, int64_t capacity)
{
    const int blocks = ((int)capacity + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(dlq_ring_append_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        slots.data_ptr<int>(), count.data_ptr<int>(), ring.data_ptr<int>(),
        head.data_ptr<int>(), (int)ring.size(0));
}

void worker_precompute_into(
    torch::Tensor w_pool, torch::Tensor w_active, torch::Tensor w_maxp,
    torch::Tensor w_cpu, torch::Tensor w_gpu, torch::Tensor out_keys)
{
    const int NW = (int)w_pool.size(0);
    if (NW == 0) return;
    const int blocks = (NW + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(worker_precompute_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        w_pool.data_ptr<int>(), w_active.data_ptr<int>(), w_maxp.data_ptr<int>(),
        w_cpu.data_ptr<float>(), w_gpu.data_ptr<float>(),
        (unsigned long long*)out_keys.data_ptr<int64_t>(), NW);
}

void spread_pick(torch::Tensor order, torch::Tensor valid_count,
                 torch::Tensor j_poolmask, torch::Tensor j_labels,
                 int64_t full_mask, torch::Tensor pick, int64_t K)
{
    const int NJ = (int)pick.size(0);
    const int blocks = (NJ + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(spread_pick_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        order.data_ptr<int>(), valid_count.data_ptr<int>(),
        (const long long*)j_poolmask.data_ptr<int64_t>(),
        (const long long*)j_labels.data_ptr<int64_t>(),
        (long long)full_mask, pick.data_ptr<int>(), NJ, (int)K);
}

void set_transition_lut(torch::Tensor lut) {
    TORCH_CHECK(lut.numel() == N_STATES * N_STATES, "lut must be 11x11");
    auto cpu = lut.to(torch::kUInt8).contiguous().cpu();
    (void)hipMemcpyToSymbol(HIP_SYMBOL(d_transition_lut), cpu.data_ptr<uint8_t>(),
                            N_STATES * N_STATES);
}

torch::Tensor apply_transitions(
    torch::Tensor states, torch::Tensor attempts, torch::Tensor deadlines,
    torch::Tensor slots, torch::Tensor to_states)
{
    CHECK_DEV(states); CHECK_DEV(slots);
    const int B = (int)slots.size(0);
    auto ok = torch::zeros({B}, torch::TensorOptions().dtype(torch::kUInt8).device(states.device()));
    if (B == 0) return ok;
    const int blocks = (B + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(apply_transitions_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        states.data_ptr<uint8_t>(), attempts.data_ptr<int>(),
        slots.data_ptr<int>(), to_states.data_ptr<uint8_t>(),
        ok.data_ptr<uint8_t>(),
        (long long*)deadlines.data_ptr<int64_t>(), B);
    return ok;
}

std::tuple<torch::Tensor, torch::Tensor> deadline_scan(
    torch::Tensor states, torch::Tensor deadlines, torch::Tensor updated_at,
    int64_t now_us, int64_t dispatch_cutoff_us, int64_t running_cutoff_us, int64_t cap)
{
    CHECK_DEV(states);
    const int N = (int)states.size(0);
    auto slots = torch::empty({cap}, torch::TensorOptions().dtype(torch::kInt32).device(states.device()));
    auto count = torch::zeros({1}, torch::TensorOptions().dtype(torch::kInt32).device(states.device()));
    if (N == 0) return {slots, count};
    const int blocks = (N + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(deadline_scan_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        states.data_ptr<uint8_t>(),
        (const long long*)deadlines.data_ptr<int64_t>(),
        (const long long*)updated_at.data_ptr<int64_t>(),
        (long long)now_us, (long long)dispatch_cutoff_us, (long long)running_cutoff_us,
        slots.data_ptr<int>(), count.data_ptr<int>(), N, (int)cap);
    return {slots, count};
}

torch::Tensor echo_execute(torch::Tensor ctx_arena, torch::Tensor res_arena, int64_t stride)
{
    CHECK_DEV(ctx_arena); CHECK_CONTIG(ctx_arena);
    const int B = (int)(ctx_arena.numel() / stride);
    auto sums = torch::empty({B}, torch::TensorOptions().dtype(torch::kInt32).device(ctx_arena.device()));
    const int waves_per_block = BLOCK / WAVE;
    const int blocks = (B + waves_per_block - 1) / waves_per_block;
    hipLaunchKernelGGL(echo_worker_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        (const unsigned int*)ctx_arena.data_ptr<int32_t>(),
        (unsigned int*)res_arena.data_ptr<int32_t>(),
        (unsigned int*)sums.data_ptr<int32_t>(), B, (int)stride);
    return sums;
}

torch::Tensor echo_execute_indexed(torch::Tensor ctx_arena, torch::Tensor slots,
                                   torch::Tensor res_arena, torch::Tensor res_sum,
                                   int64_t stride)
{
    CHECK_DEV(ctx_arena); CHECK_DEV(slots);
    const int B = (int)slots.size(0);
    if (B == 0) return res_sum;
    const int waves_per_block = BLOCK / WAVE;
    const int blocks = (B + waves_per_block - 1) / waves_per_block;
    hipLaunchKernelGGL(echo_worker_indexed_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        (const unsigned int*)ctx_arena.data_ptr<int32_t>(),
        slots.data_ptr<int>(),
        (unsigned int*)res_arena.data_ptr<int32_t>(),
        (unsigned int*)res_sum.data_ptr<int32_t>(), B, (int)stride);
    return res_sum;
}

torch::Tensor policy_first_match_mfma(
    torch::Tensor a_pack, torch::Tensor b_pack, torch::Tensor cards,
    torch::Tensor rule_secrets, torch::Tensor job_secrets,
    torch::Tensor tile_dims, int64_t n_jobs, int64_t n_rules)
{
    CHECK_DEV(a_pack); CHECK_DEV(b_pack);
    const int J = (int)n_jobs, R = (int)n_rules;
    const int Jt = (int)a_pack.size(0);
    const int Rt = (R + 15) / 16;
    auto out = torch::full({J}, INT_MAX,
        torch::TensorOptions().dtype(torch::kInt32).device(a_pack.device()));
    if (J == 0 || R == 0) return out.masked_fill_(out == INT_MAX, -1);
    // fill the chip: >=2048 workgroups via rule chunking
    int nchunks = std::max(1, std::min((Rt + 3) / 4, std::max(1, 2048 / std::max(Jt, 1))));
    const int tiles_per_chunk = (Rt + nchunks - 1) / nchunks;
    hipLaunchKernelGGL(policy_first_match_mfma_kernel, dim3(Jt, nchunks), dim3(BLOCK), 0, cur_stream(),
        (const signed char*)a_pack.data_ptr<int8_t>(),
        (const signed char*)b_pack.data_ptr<int8_t>(),
        cards.data_ptr<int>(),
        (const signed char*)rule_secrets.data_ptr<int8_t>(),
        job_secrets.data_ptr<uint8_t>(),
        tile_dims.data_ptr<int>(),
        out.data_ptr<int>(), J, R, tiles_per_chunk);
    out.masked_fill_(out == INT_MAX, -1);
    return out;
}

void policy_gate(torch::Tensor first, torch::Tensor decisions, torch::Tensor out_decision,
                 torch::Tensor denied_slots, torch::Tensor denied_count,
                 torch::Tensor allowed_slots, torch::Tensor allowed_count)
{
    const int J = (int)first.size(0);
    const int blocks = (J + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(policy_gate_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        first.data_ptr<int>(), (const signed char*)decisions.data_ptr<int8_t>(),
        (signed char*)out_decision.data_ptr<int8_t>(),
        denied_slots.data_ptr<int>(), denied_count.data_ptr<int>(),
        allowed_slots.data_ptr<int>(), allowed_count.data_ptr<int>(), J);
}

void compact_routable(torch::Tensor allowed_slots, torch::Tensor allowed_count,
                      torch::Tensor pick, torch::Tensor routable_slots,
                      torch::Tensor routable_widx, torch::Tensor routable_count)
{
    const int J = (int)pick.size(0);
    const int blocks = (J + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(compact_routable_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        allowed_slots.data_ptr<int>(), allowed_count.data_ptr<int>(),
        pick.data_ptr<int>(), routable_slots.data_ptr<int>(),
        routable_widx.data_ptr<int>(), routable_count.data_ptr<int>());
}

void policy_gate_full(torch::Tensor first, torch::Tensor decisions, torch::Tensor out_decision,
                      torch::Tensor denied_slots, torch::Tensor denied_count,
                      torch::Tensor allowed_slots, torch::Tensor allowed_count,
                      torch::Tensor states, torch::Tensor deadlines,
                      torch::Tensor dlq_ring, torch::Tensor dlq_head)
{
    const int J = (int)first.size(0);
    const int blocks = (J + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(policy_gate_full_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        first.data_ptr<int>(), (const signed char*)decisions.data_ptr<int8_t>(),
        (signed char*)out_decision.data_ptr<int8_t>(),
        denied_slots.data_ptr<int>(), denied_count.data_ptr<int>(),
        allowed_slots.data_ptr<int>(), allowed_count.data_ptr<int>(),
        states.data_ptr<uint8_t>(), (long long*)deadlines.data_ptr<int64_t>(),
        dlq_ring.data_ptr<int>(), dlq_head.data_ptr<int>(),
        (int)dlq_ring.size(0), J, (int)decisions.size(0));
}
void tick_reset(torch::Tensor states, torch::Tensor counts)
{
    const int B = (int)states.size(0);
    const int n = (int)counts.size(0);
    const int blocks = (std::max(B, n) + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(tick_reset_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        states.data_ptr<uint8_t>(), B, counts.data_ptr<int>(), n);
}
void accumulate_counts(torch::Tensor counts, torch::Tensor acc)
{
    const int n = (int)counts.size(0);
    hipLaunchKernelGGL(accumulate_counts_kernel, dim3(1), dim3(WAVE), 0, cur_stream(),
        counts.data_ptr<int>(), (long long*)acc.data_ptr<int64_t>(), n);
}
void begin_tick(torch::Tensor states, torch::Tensor counts)
{
    const int B = (int)states.size(0);
    const int n = (int)counts.size(0);
    const int blocks = (std::max(B, n) + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(begin_tick_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        states.data_ptr<uint8_t>(), B, counts.data_ptr<int>(), n, (int*)nullptr);
}
void begin_tick_first(torch::Tensor states, torch::Tensor counts, torch::Tensor first)
{
    const int B = (int)states.size(0);
    const int n = (int)counts.size(0);
    const int blocks = (std::max(B, n) + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(begin_tick_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        states.data_ptr<uint8_t>(), B, counts.data_ptr<int>(), n, first.data_ptr<int>());
}
void policy_first_match_mfma_into(
    torch::Tensor a_pack, torch::Tensor b_pack, torch::Tensor cards,
    torch::Tensor rule_secrets, torch::Tensor job_secrets,
    torch::Tensor tile_dims, int64_t n_jobs, int64_t n_rules, torch::Tensor out)
{
    CHECK_DEV(a_pack); CHECK_DEV(b_pack);
    const int J = (int)n_jobs, R = (int)n_rules;
    const int Jt = (int)a_pack.size(0);
    const int Rt = (R + 15) / 16;
    if (J == 0 || R == 0) return;
    int nchunks = std::max(1, std::min((Rt + 3) / 4, std::max(1, 2048 / std::max(Jt, 1))));
    const int tiles_per_chunk = (Rt + nchunks - 1) / nchunks;
    hipLaunchKernelGGL(policy_first_match_mfma_kernel, dim3(Jt, nchunks), dim3(BLOCK), 0, cur_stream(),
        (const signed char*)a_pack.data_ptr<int8_t>(),
        (const signed char*)b_pack.data_ptr<int8_t>(),
        cards.data_ptr<int>(),
        (const signed char*)rule_secrets.data_ptr<int8_t>(),
        job_secrets.data_ptr<uint8_t>(),
        tile_dims.data_ptr<int>(),
        out.data_ptr<int>(), J, R, tiles_per_chunk);
}
void compact_routable_spread(torch::Tensor allowed_slots, torch::Tensor allowed_count,
                             torch::Tensor pick, torch::Tensor order, torch::Tensor valid_count,
                             torch::Tensor j_poolmask, torch::Tensor j_labels,
                             int64_t full_mask, int64_t K,
                             torch::Tensor routable_slots, torch::Tensor routable_widx,
                             torch::Tensor routable_count)
{
    const int J = (int)allowed_slots.size(0);
    const int blocks = (J + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(compact_routable_spread_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        allowed_slots.data_ptr<int>(), allowed_count.data_ptr<int>(),
        pick.data_ptr<int>(), order.data_ptr<int>(), valid_count.data_ptr<int>(),
        (const long long*)j_poolmask.data_ptr<int64_t>(),
        (const long long*)j_labels.data_ptr<int64_t>(),
        (long long)full_mask, (int)K,
        routable_slots.data_ptr<int>(), routable_widx.data_ptr<int>(),
        routable_count.data_ptr<int>());
}
void apply_transitions_chain_dyn(torch::Tensor states, torch::Tensor attempts,
                                 torch::Tensor deadlines, torch::Tensor slots,
                                 torch::Tensor count, std::vector<int64_t> chain,
                                 torch::Tensor extra_zero, int64_t capacity)
{
    int to[4] = {-1, -1, -1, -1};
    for (size_t c = 0; c < chain.size() && c < 4; ++c) to[c] = (int)chain[c];
    const int n_extra = (int)extra_zero.size(0);
    const int blocks = (std::max((int)capacity, n_extra) + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(apply_transitions_chain_dyn_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        states.data_ptr<uint8_t>(), attempts.data_ptr<int>(),
        (long long*)deadlines.data_ptr<int64_t>(),
        slots.data_ptr<int>(), count.data_ptr<int>(),
        to[0], to[1], to[2], to[3],
        extra_zero.data_ptr<int>(), n_extra);
}
void apply_transitions_dyn(torch::Tensor states, torch::Tensor attempts, torch::Tensor deadlines,
                           torch::Tensor slots, torch::Tensor count, int64_t to_state, int64_t capacity)
{
    const int blocks = ((int)capacity + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(apply_transitions_dyn_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        states.data_ptr<uint8_t>(), attempts.data_ptr<int>(),
        (long long*)deadlines.data_ptr<int64_t>(),
        slots.data_ptr<int>(), count.data_ptr<int>(), (unsigned char)to_state);
}

void echo_execute_indexed_dyn(torch::Tensor ctx_arena, torch::Tensor slots, torch::Tensor count,
                              torch::Tensor res_arena, torch::Tensor res_sum,
                              int64_t stride, int64_t capacity)
{
    const int waves_per_block = BLOCK / WAVE;
    const int blocks = ((int)capacity + waves_per_block - 1) / waves_per_block;
    hipLaunchKernelGGL(echo_worker_indexed_dyn_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        (const unsigned int*)ctx_arena.data_ptr<int32_t>(), slots.data_ptr<int>(),
        count.data_ptr<int>(), (unsigned int*)res_arena.data_ptr<int32_t>(),
        (unsigned int*)res_sum.data_ptr<int32_t>(), (int)stride);
}

void load_feedback(torch::Tensor routable_widx, torch::Tensor count,
                   torch::Tensor w_active_local, int64_t nwl, int64_t my_rank, int64_t capacity)
{
    const int blocks = ((int)capacity + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(load_feedback_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        routable_widx.data_ptr<int>(), count.data_ptr<int>(),
        w_active_local.data_ptr<int>(), (int)nwl, (int)my_rank);
}

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor, torch::Tensor> run_readiness(
    torch::Tensor step_state, torch::Tensor deps_mask, torch::Tensor n_steps,
    torch::Tensor run_active, int64_t cap)
{
    CHECK_DEV(step_state);
    const int NR = (int)step_state.size(0);
    auto opts_i = torch::TensorOptions().dtype(torch::kInt32).device(step_state.device());
    auto ready = torch::zeros({NR}, torch::TensorOptions().dtype(torch::kInt64).device(step_state.device()));
    auto runs = torch::empty({cap}, opts_i);
    auto steps = torch::empty({cap}, opts_i);
    auto count = torch::zeros({1}, opts_i);
    if (NR == 0) return {ready, runs, steps, count};
    const int blocks = (NR + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(run_readiness_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        step_state.data_ptr<uint8_t>(),
        (const long long*)deps_mask.data_ptr<int64_t>(),
        n_steps.data_ptr<uint8_t>(), run_active.data_ptr<uint8_t>(),
        (long long*)ready.data_ptr<int64_t>(),
        runs.data_ptr<int>(), steps.data_ptr<int>(), count.data_ptr<int>(),
        NR, (int)cap);
    return {ready, runs, steps, count};
}


void pack_by_dest(torch::Tensor routable_slots, torch::Tensor routable_widx,
                  torch::Tensor routable_count, torch::Tensor send_slots,
                  torch::Tensor send_widx, torch::Tensor send_cnt,
                  int64_t nwl, int64_t cap, int64_t capacity,
                  torch::Tensor rq_src, torch::Tensor rq_widx,
                  torch::Tensor rq_attempts, torch::Tensor rq_count,
                  torch::Tensor rq_dead,
                  torch::Tensor dead_src, torch::Tensor dead_count)
{
    const int blocks = ((int)capacity + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(pack_by_dest_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        routable_slots.data_ptr<int>(), routable_widx.data_ptr<int>(),
        routable_count.data_ptr<int>(), send_slots.data_ptr<int>(),
        send_widx.data_ptr<int>(), send_cnt.data_ptr<int>(), (int)nwl, (int)cap,
        rq_src.data_ptr<int>(), rq_widx.data_ptr<int>(),
        rq_attempts.data_ptr<int>(), rq_count.data_ptr<int>(),
        (unsigned long long*)rq_dead.data_ptr<int64_t>(),
        (int)rq_src.size(0),
        dead_src.data_ptr<int>(), dead_count.data_ptr<int>(),
        (int)dead_src.size(0));
}

void pack_requeue(torch::Tensor rq_prev_widx, torch::Tensor rq_prev_attempts,
                  torch::Tensor rq_prev_count, torch::Tensor send_slots,
                  torch::Tensor send_widx, torch::Tensor send_cnt,
                  int64_t nwl, int64_t cap,
                  torch::Tensor rq_src, torch::Tensor rq_widx,
                  torch::Tensor rq_attempts, torch::Tensor rq_count,
                  torch::Tensor rq_dead,
                  torch::Tensor dead_src, torch::Tensor dead_count)
{
    const int rq_cap = (int)rq_src.size(0);
    const int blocks = (rq_cap + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(pack_requeue_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        rq_prev_widx.data_ptr<int>(), rq_prev_attempts.data_ptr<int>(),
        rq_prev_count.data_ptr<int>(), send_slots.data_ptr<int>(),
        send_widx.data_ptr<int>(), send_cnt.data_ptr<int>(), (int)nwl, (int)cap,
        rq_src.data_ptr<int>(), rq_widx.data_ptr<int>(),
        rq_attempts.data_ptr<int>(), rq_count.data_ptr<int>(),
        (unsigned long long*)rq_dead.data_ptr<int64_t>(), rq_cap,
        dead_src.data_ptr<int>(), dead_count.data_ptr<int>(),
        (int)dead_src.size(0));
}

void wf_sweep(torch::Tensor step_state, torch::Tensor deps_mask, torch::Tensor n_steps,
              torch::Tensor run_active, torch::Tensor step_kind, torch::Tensor cond_bits,
              torch::Tensor next_ready, torch::Tensor tick,
              torch::Tensor disp_runs, torch::Tensor disp_steps, torch::Tensor disp_count,
              torch::Tensor appr_runs, torch::Tensor appr_steps, torch::Tensor appr_count)
{
    const int NR = (int)n_steps.size(0);
    const int blocks = (NR + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(wf_sweep_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        step_state.data_ptr<uint8_t>(), (const long long*)deps_mask.data_ptr<int64_t>(),
        n_steps.data_ptr<uint8_t>(), run_active.data_ptr<uint8_t>(),
        step_kind.data_ptr<uint8_t>(), (const long long*)cond_bits.data_ptr<int64_t>(),
        next_ready.data_ptr<int>(), tick.data_ptr<int>(),
        disp_runs.data_ptr<int>(), disp_steps.data_ptr<int>(), disp_count.data_ptr<int>(),
        appr_runs.data_ptr<int>(), appr_steps.data_ptr<int>(), appr_count.data_ptr<int>(),
        NR, (int)disp_runs.size(0), (int)appr_runs.size(0));
}

void wf_expand(torch::Tensor disp_runs, torch::Tensor disp_steps, torch::Tensor disp_count,
               torch::Tensor step_state, torch::Tensor children_todo, torch::Tensor children_out,
               torch::Tensor max_parallel,
               torch::Tensor child_tag, torch::Tensor child_seq, torch::Tensor child_widx,
               torch::Tensor child_count, torch::Tensor children_emitted,
               torch::Tensor dispatch_tick, torch::Tensor tick,
               torch::Tensor order, torch::Tensor valid_count)
{
    const int cap = (int)disp_runs.size(0);
    const int blocks = (cap + (BLOCK / WAVE) - 1) / (BLOCK / WAVE);
    hipLaunchKernelGGL(wf_expand_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        disp_runs.data_ptr<int>(), disp_steps.data_ptr<int>(), disp_count.data_ptr<int>(),
        step_state.data_ptr<uint8_t>(), children_todo.data_ptr<int>(),
        children_out.data_ptr<int>(), max_parallel.data_ptr<int>(), child_tag.data_ptr<int>(),
        child_seq.data_ptr<int>(), child_widx.data_ptr<int>(), child_count.data_ptr<int>(),
        children_emitted.data_ptr<int>(),
        dispatch_tick.data_ptr<int>(), tick.data_ptr<int>(),
        order.data_ptr<int>(), valid_count.data_ptr<int>(),
        cap, (int)child_tag.size(0));
}

void wf_timeout_scan(torch::Tensor step_state, torch::Tensor children_out,
                     torch::Tensor children_fail, torch::Tensor dispatch_tick,
                     torch::Tensor tick, int64_t cutoff, torch::Tensor timeout_count)
{
    const int NRS = (int)step_state.numel();
    const int blocks = (NRS + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(wf_timeout_scan_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        step_state.data_ptr<uint8_t>(), children_out.data_ptr<int>(),
        children_fail.data_ptr<int>(), dispatch_tick.data_ptr<int>(),
        tick.data_ptr<int>(), (int)cutoff,
        (unsigned long long*)timeout_count.data_ptr<int64_t>(), NRS);
}

void wf_readmit(torch::Tensor run_active, torch::Tensor run_state, torch::Tensor n_steps,
                torch::Tensor step_state, torch::Tensor step_attempts,
                torch::Tensor children_todo, torch::Tensor children_out,
                torch::Tensor children_done, torch::Tensor children_fail,
                torch::Tensor children_emitted, torch::Tensor next_ready,
                torch::Tensor dispatch_tick, torch::Tensor todo_tmpl,
                torch::Tensor nready_tmpl, int64_t filter_state,
                torch::Tensor admit_count)
{
    const int NR = (int)n_steps.size(0);
    const int blocks = (NR + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(wf_readmit_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        run_active.data_ptr<uint8_t>(), run_state.data_ptr<uint8_t>(),
        n_steps.data_ptr<uint8_t>(), step_state.data_ptr<uint8_t>(),
        step_attempts.data_ptr<int>(), children_todo.data_ptr<int>(),
        children_out.data_ptr<int>(), children_done.data_ptr<int>(),
        children_fail.data_ptr<int>(), children_emitted.data_ptr<int>(),
        next_ready.data_ptr<int>(), dispatch_tick.data_ptr<int>(),
        todo_tmpl.data_ptr<int>(), nready_tmpl.data_ptr<int>(),
        (int)filter_state,
        (unsigned long long*)admit_count.data_ptr<int64_t>(), NR);
}

void wf_apply(torch::Tensor send_slots, torch::Tensor send_cnt, torch::Tensor child_tag,
              torch::Tensor child_seq, torch::Tensor rq_prev_tag, torch::Tensor rq_prev_seq,
              torch::Tensor children_done,
              torch::Tensor children_fail, torch::Tensor children_out,
              int64_t fail_ppt, int64_t drop_ppt, int64_t cap, int64_t world)
{
    const int n = (int)(world * cap);
    const int blocks = (n + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(wf_apply_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        send_slots.data_ptr<int>(), send_cnt.data_ptr<int>(), child_tag.data_ptr<int>(),
        child_seq.data_ptr<int>(), rq_prev_tag.data_ptr<int>(), rq_prev_seq.data_ptr<int>(),
        children_done.data_ptr<int>(),
        children_fail.data_ptr<int>(), children_out.data_ptr<int>(),
        (int)fail_ppt, (int)drop_ppt, (int)cap, (int)world);
}

void wf_apply_dead(torch::Tensor dead_src, torch::Tensor dead_count, torch::Tensor child_tag,
                   torch::Tensor rq_prev_tag, torch::Tensor children_fail,
                   torch::Tensor children_out)
{
    const int dcap = (int)dead_src.size(0);
    const int blocks = (dcap + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(wf_apply_dead_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        dead_src.data_ptr<int>(), dead_count.data_ptr<int>(), child_tag.data_ptr<int>(),
        rq_prev_tag.data_ptr<int>(), children_fail.data_ptr<int>(),
        children_out.data_ptr<int>(), dcap);
}

void wf_commit(torch::Tensor step_state, torch::Tensor step_attempts,
               torch::Tensor children_todo, torch::Tensor children_out,
               torch::Tensor children_done, torch::Tensor children_fail,
               torch::Tensor max_parallel,
               torch::Tensor next_ready, torch::Tensor tick, int64_t max_retries,
               torch::Tensor retry_count)
{
    const int NRS = (int)step_state.numel();
    const int blocks = (NRS + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(wf_commit_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        step_state.data_ptr<uint8_t>(), step_attempts.data_ptr<int>(),
        children_todo.data_ptr<int>(), children_out.data_ptr<int>(),
        children_done.data_ptr<int>(), children_fail.data_ptr<int>(),
        max_parallel.data_ptr<int>(),
        next_ready.data_ptr<int>(), tick.data_ptr<int>(), (int)max_retries,
        (unsigned long long*)retry_count.data_ptr<int64_t>(), NRS);
}

void wf_status(torch::Tensor step_state, torch::Tensor n_steps, torch::Tensor run_active,
               torch::Tensor run_state, torch::Tensor counts)
{
    const int NR = (int)n_steps.size(0);
    const int blocks = (NR + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(wf_status_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        step_state.data_ptr<uint8_t>(), n_steps.data_ptr<uint8_t>(),
        run_active.data_ptr<uint8_t>(), run_state.data_ptr<uint8_t>(),
        (unsigned long long*)counts.data_ptr<int64_t>(), NR);
}

void wf_grant(torch::Tensor grant_runs, torch::Tensor grant_steps, torch::Tensor verdicts,
              int64_t n, torch::Tensor step_state)
{
    if (n <= 0) return;
    const int blocks = ((int)n + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(wf_grant_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        grant_runs.data_ptr<int>(), grant_steps.data_ptr<int>(),
        verdicts.data_ptr<uint8_t>(), (int)n, step_state.data_ptr<uint8_t>());
}

void materialize_rq_payload(torch::Tensor payload, torch::Tensor rq_prev_payload,
                            torch::Tensor rq_src, torch::Tensor rq_count,
                            torch::Tensor rq_payload, int64_t stride)
{
    const int rq_cap = (int)rq_src.size(0);
    const int blocks = (rq_cap + (BLOCK / WAVE) - 1) / (BLOCK / WAVE);
    hipLaunchKernelGGL(materialize_rq_payload_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        (const unsigned int*)payload.data_ptr<int32_t>(),
        (const unsigned int*)rq_prev_payload.data_ptr<int32_t>(),
        rq_src.data_ptr<int>(), rq_count.data_ptr<int>(),
        (unsigned int*)rq_payload.data_ptr<int32_t>(), (int)stride, rq_cap);
}

void gather_payload_padded(torch::Tensor payload, torch::Tensor rq_prev_payload,
                           torch::Tensor send_slots,
                           torch::Tensor send_cnt, torch::Tensor send_payload,
                           int64_t stride, int64_t cap, int64_t world)
{
    const int waves = (int)(world * cap);
    const int blocks = (waves + (BLOCK / WAVE) - 1) / (BLOCK / WAVE);
    hipLaunchKernelGGL(gather_payload_padded_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        (const unsigned int*)payload.data_ptr<int32_t>(),
        (const unsigned int*)rq_prev_payload.data_ptr<int32_t>(),
        send_slots.data_ptr<int>(),
        send_cnt.data_ptr<int>(), (unsigned int*)send_payload.data_ptr<int32_t>(),
        (int)stride, (int)cap, (int)world);
}

void echo_padded(torch::Tensor recv_payload, torch::Tensor recv_cnt,
                 torch::Tensor res_arena, torch::Tensor res_sums,
                 int64_t stride, int64_t cap, int64_t world)
{
    const int waves = (int)(world * cap);
    const int blocks = (waves + (BLOCK / WAVE) - 1) / (BLOCK / WAVE);
    hipLaunchKernelGGL(echo_padded_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        (const unsigned int*)recv_payload.data_ptr<int32_t>(), recv_cnt.data_ptr<int>(),
        (unsigned int*)res_arena.data_ptr<int32_t>(),
        (unsigned int*)res_sums.data_ptr<int32_t>(), (int)stride, (int)cap, (int)world);
}

void apply_transitions_padded(torch::Tensor states, torch::Tensor attempts,
                              torch::Tensor deadlines, torch::Tensor slots_pad,
                              torch::Tensor cnt, int64_t to_state, int64_t cap, int64_t world)
{
    const int n = (int)(world * cap);
    const int blocks = (n + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(apply_transitions_padded_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        states.data_ptr<uint8_t>(), attempts.data_ptr<int>(),
        (long long*)deadlines.data_ptr<int64_t>(), slots_pad.data_ptr<int>(),
        cnt.data_ptr<int>(), (unsigned char)to_state, (int)cap, (int)world);
}

void load_feedback_padded(torch::Tensor recv_widx, torch::Tensor recv_cnt,
                          torch::Tensor w_active_local, int64_t cap, int64_t world)
{
    const int n = (int)(world * cap);
    const int blocks = (n + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(load_feedback_padded_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        recv_widx.data_ptr<int>(), recv_cnt.data_ptr<int>(),
        w_active_local.data_ptr<int>(), (int)cap, (int)world);
}


// ---------------------------------------------------------------------------
// Host-side fused synthetic-batch encoder (the e2e ingest stand-in).
// The torch-op version (SyntheticEncoder.fresh) is RNG+gather bound at
// ~1 ms/16k batch; this fused pass (counter-based splitmix64 RNG + LUT row
// gather) does the same work at memory speed. It runs SERIAL and releases
// the GIL: the e2e loop overlaps 2-3 encoder threads, each a fully
// independent call — a shared at::parallel_for pool serialized those calls
// (and holding the GIL stalled the submit loop for the whole encode).
// Index draws use Lemire multiply-shift reduction ((u32)h * V >> 32) —
// a plain `% V` was ~half the per-job cost.
// Deterministic: draws depend only on (seed, step, job, stream), so every
// backend and every re-run produces identical batches.
// ---------------------------------------------------------------------------
static inline uint64_t splitmix64(uint64_t x) {
    x += 0x9E3779B97F4A7C15ull;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
    return x ^ (x >> 31);
}

void synthetic_fresh(torch::Tensor any_bits,   // [B,7,W] int64 (host)
                     torch::Tensor all_bits,   // [B,2,W] int64 (host)
                     torch::Tensor tenant_lut, // [V,W] int64
                     torch::Tensor topic_lut,  // [V,W]
                     torch::Tensor risk_lut,   // [V,W]
                     torch::Tensor req_lut,    // [V,W]
                     int64_t seed, int64_t step,
                     int64_t dim_tenant, int64_t dim_topic, int64_t dim_risk,
                     int64_t all_requires)
{
    TORCH_CHECK(!any_bits.is_cuda(), "synthetic_fresh is a host encoder");
    const int64_t B = any_bits.size(0);
    const int64_t W = any_bits.size(2);
    const int64_t V = tenant_lut.size(0);
    int64_t* any = any_bits.data_ptr<int64_t>();
    int64_t* all = all_bits.data_ptr<int64_t>();
    const int64_t* tl = tenant_lut.data_ptr<int64_t>();
    const int64_t* ol = topic_lut.data_ptr<int64_t>();
    const int64_t* rl = risk_lut.data_ptr<int64_t>();
    const int64_t* ql = req_lut.data_ptr<int64_t>();
    const uint64_t base = splitmix64((uint64_t)seed * 0x5851F42D4C957F2Dull
                                     ^ (uint64_t)step);
    const uint64_t v = (uint64_t)V;
    constexpr uint32_t RISK_CUT = 1288490189u;  // 0.30 * 2^32
    constexpr uint32_t REQ_CUT = 429496730u;    // 0.10 * 2^32
    if (W == 1) {
        // fast path for the e2e bench shape (single bitset word per dim)
        for (int64_t i = 0; i < B; ++i) {
            const uint64_t h0 = splitmix64(base ^ (uint64_t)i);
            const uint64_t h1 = splitmix64(h0);
            const uint64_t h2 = splitmix64(h1);
            const uint64_t t_idx = ((h0 & 0xffffffffull) * v) >> 32;
            const uint64_t o_idx = ((h0 >> 32) * v) >> 32;
            const uint64_t r_idx = ((h1 & 0xffffffffull) * v) >> 32;
            const uint64_t q_idx = ((h1 >> 32) * v) >> 32;
            const bool has_risk = (uint32_t)h2 < RISK_CUT;
            const bool has_req = (uint32_t)(h2 >> 32) < REQ_CUT;
            int64_t* arow = any + (size_t)i * 7;
            arow[dim_tenant] = tl[t_idx];
            arow[dim_topic] = ol[o_idx];
            arow[dim_risk] = has_risk ? rl[r_idx] : 0;
            all[(size_t)i * 2 + all_requires] = has_req ? ql[q_idx] : 0;
        }
        return;
    }
    for (int64_t i = 0; i < B; ++i) {
        const uint64_t h0 = splitmix64(base ^ (uint64_t)i);
        const uint64_t h1 = splitmix64(h0);
        const uint64_t h2 = splitmix64(h1);
        const int64_t t_idx = (int64_t)(((h0 & 0xffffffffull) * v) >> 32);
        const int64_t o_idx = (int64_t)(((h0 >> 32) * v) >> 32);
        const int64_t r_idx = (int64_t)(((h1 & 0xffffffffull) * v) >> 32);
        const int64_t q_idx = (int64_t)(((h1 >> 32) * v) >> 32);
        const bool has_risk = (uint32_t)h2 < RISK_CUT;
        const bool has_req = (uint32_t)(h2 >> 32) < REQ_CUT;
        int64_t* arow = any + (size_t)i * 7 * W;
        int64_t* lrow = all + (size_t)i * 2 * W;
        for (int64_t w = 0; w < W; ++w) {
            arow[dim_tenant * W + w] = tl[t_idx * W + w];
            arow[dim_topic * W + w] = ol[o_idx * W + w];
            arow[dim_risk * W + w] = has_risk ? rl[r_idx * W + w] : 0;
            lrow[all_requires * W + w] = has_req ? ql[q_idx * W + w] : 0;
        }
    }
}


void pack_jobs_mfma_dev(torch::Tensor any_bits, torch::Tensor all_bits,
                        torch::Tensor a_pack)
{
    const int B = (int)any_bits.size(0);
    const int Jt = (int)a_pack.size(0);
    const int n = Jt * 9 * 64;
    const int blocks = (n + BLOCK - 1) / BLOCK;
    hipLaunchKernelGGL(pack_jobs_mfma_kernel, dim3(blocks), dim3(BLOCK), 0, cur_stream(),
        (const long long*)any_bits.data_ptr<int64_t>(),
        (const long long*)all_bits.data_ptr<int64_t>(),
        a_pack.data_ptr<int8_t>(), B, Jt);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("policy_first_match", &policy_first_match, "K1 batched policy first-match");
    m.def("policy_first_match_mfma", &policy_first_match_mfma, "K1 MFMA comparison variant");
    m.def("worker_precompute", &worker_precompute, "K2a per-worker score/overload precompute");
    m.def("least_loaded_pick", &least_loaded_pick, "K2 least-loaded worker pick");
    m.def("spread_pick", &spread_pick, "K2c batch spreading over the least-loaded set");
    m.def("worker_precompute_into", &worker_precompute_into, "K2a into a persistent key tensor");
    m.def("dlq_ring_append", &dlq_ring_append, "K7 device DLQ ring append");
    m.def("echo_execute_indexed", &echo_execute_indexed, "device echo worker pool (slot-indexed, in-place)");
    m.def("policy_gate", &policy_gate, "decision gather + allow/deny compaction");
    m.def("policy_gate_full", &policy_gate_full,
          "gate + DENIED transition + DLQ ring append in one launch");
    m.def("tick_reset", &tick_reset, "one-launch per-tick state/counter reset");
    m.def("begin_tick", &begin_tick, "tick prologue: PENDING re-admit + counter reset");
    m.def("accumulate_counts", &accumulate_counts, "fold tick counters into device accumulator");
    m.def("begin_tick_first", &begin_tick_first,
          "tick prologue that also primes the first-match buffer (INT_MAX)");
    m.def("least_loaded_pick_into", &least_loaded_pick_into,
          "K2 pick into a persistent buffer, skipping unconstrained jobs");
    m.def("policy_first_match_mfma_into", &policy_first_match_mfma_into,
          "K1 MFMA into a pre-primed buffer (no alloc/fill launches)");
    m.def("compact_routable_spread", &compact_routable_spread,
          "routable compaction with inline K2c spread");
    m.def("apply_transitions_chain_dyn", &apply_transitions_chain_dyn,
          "K5 chained targets in one launch (+ extra zero region)");
    m.def("compact_routable", &compact_routable, "routable-slot compaction");
    m.def("apply_transitions_dyn", &apply_transitions_dyn, "K5 with device-resident count");
    m.def("echo_execute_indexed_dyn", &echo_execute_indexed_dyn, "echo worker with device count");
    m.def("load_feedback", &load_feedback, "per-worker active-count histogram");
    m.def("run_readiness", &run_readiness, "K3 run/step readiness sweep");
    // GIL released: encoder threads run concurrently with the submit loop
    m.def("synthetic_fresh", &synthetic_fresh,
          py::call_guard<py::gil_scoped_release>(),
          "fused host synthetic-batch encoder");
    m.def("pack_jobs_mfma_dev", &pack_jobs_mfma_dev, "device job packing for the MFMA K1");
    m.def("pack_by_dest", &pack_by_dest, "padded per-destination dispatch pack");
    m.def("pack_requeue", &pack_requeue, "redeliver last tick's requeue ring into the send segments");
    m.def("wf_sweep", &wf_sweep, "K3-WF readiness sweep + approval holds + condition/delay commits");
    m.def("wf_expand", &wf_expand, "K3-WF for_each/worker child expansion into the child arena");
    m.def("wf_apply", &wf_apply, "K3-WF child result application with failure injection");
    m.def("wf_apply_dead", &wf_apply_dead, "K3-WF dead-letter child application");
    m.def("wf_commit", &wf_commit, "K3-WF step aggregation + retry/backoff commit");
    m.def("wf_timeout_scan", &wf_timeout_scan, "K4-WF stale-step timeout scan");
    m.def("wf_readmit", &wf_readmit, "K3-WF continuous run re-admission from template");
    m.def("wf_status", &wf_status, "K3-WF run status roll-up");
    m.def("wf_grant", &wf_grant, "K3-WF host approval grants");
    m.def("materialize_rq_payload", &materialize_rq_payload, "copy requeued payload rows into the rq arena");
    m.def("gather_payload_padded", &gather_payload_padded, "payload gather into padded send arena");
    m.def("echo_padded", &echo_padded, "region-valid echo over padded recv arena");
    m.def("apply_transitions_padded", &apply_transitions_padded, "K5 over padded slot list");
    m.def("load_feedback_padded", &load_feedback_padded, "padded per-worker load histogram");
    m.def("set_transition_lut", &set_transition_lut, "upload transition legality LUT");
    m.def("apply_transitions", &apply_transitions, "K5 batched state transitions");
    m.def("deadline_scan", &deadline_scan, "K4 deadline/staleness scan");
    m.def("echo_execute", &echo_execute, "device echo worker pool");
}
