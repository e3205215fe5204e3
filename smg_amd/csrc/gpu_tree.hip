// GPU-resident paged prefix index for cache-aware routing — gfx950 (MI355X).
//
// Re-designs the reference's host radix tree (crates/kv_index/src/token_tree.rs:303,
// match_prefix_with_counts :620, match_and_insert :754) as a device-resident
// structure sized for 288 GB HBM3E:
//
//   * one entry per (prefix depth, page), keyed by a CHAIN HASH — a rolling
//     polynomial over the whole token prefix (pages weighted by W^page).
//     The key is a pure function of the tokens, NOT of tree state, so a
//     request's 64 page lookups are INDEPENDENT: one wave matches 64 prefix
//     depths in parallel (lane p probes depth p) instead of a serial
//     parent->child walk with a ~900-cycle HBM dependency per level.
//     matched length = count of trailing ones in the presence ballot;
//   * no stored tokens are verified: the 64-bit chained key commits to the
//     entire prefix; a collision is ~2^-64 and costs one sub-optimally
//     routed request, never correctness (routing is a hint);
//   * all shared state (table, tenant bitmasks, LRU stamps, load counters)
//     is touched with agent-scope atomics which bypass the non-coherent L1s
//     (MI355X_MICROARCH.md §Workgroup dispatch: atomics/sc1 are L2-served);
//   * per-(entry, tenant) LRU stamps from a device logical clock drive the
//     eviction sweep; tenants are worker slots 0..63 (bitmask u64).
//
// The host C++ twin (host_tree.cpp) computes the IDENTICAL chain keys
// sequentially, so host and device indexes are differentially testable.
//
// Kernels: match-decide-insert (the whole cache_aware decision per batch in
// one launch), tenant removal, LRU sweep, stats.
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>

#define WAVE 64
#define EMPTY_KEY 0ull
#define TOMBSTONE_KEY 1ull

// ---------------------------------------------------------------------------
// device-side structure (all pointers into one device allocation)
// ---------------------------------------------------------------------------
struct GpuTreeDev {
    // hash table: capacity `table_size` (power of two)
    unsigned long long* table_keys;  // 0 empty, 1 tombstone, else key
    uint32_t* table_vals;            // node id
    // node pool: capacity `node_cap`
    unsigned long long* node_tenants;  // bitmask of worker slots
    uint32_t* node_ts;                 // [node_cap * 64] per-tenant LRU stamp
    uint32_t* node_parent;             // parent node id (eviction bookkeeping)
    uint32_t* node_slot;               // this node's table slot (0xffffffff = unpublished)
    // node reclamation: LIFO free list refilled by smg_tree_reclaim so a
    // long-lived gateway keeps learning after the bump allocator hits
    // node_cap (reference: token_tree.rs LRU eviction to --max-tree-size)
    uint32_t* free_list;  // [node_cap] reclaimed node ids
    int* free_top;        // stack top (count of free ids)
    // counters
    uint32_t* next_node;  // bump allocator (node 0 = root, never allocated)
    uint32_t* clock_;     // logical LRU clock
    uint32_t node_cap;
    uint32_t table_mask;  // table_size - 1
    uint32_t page_size;
    uint32_t max_pages;   // walk depth cap
};

#define NODE_SLOT_NONE 0xffffffffu

// Allocate a node id: pop the free list first, else bump-allocate.  All tree
// kernels run on ONE stream (GpuTreeHost::stream), so reclamation is never
// concurrent with match/insert — the free list needs no ABA protection.
__device__ __forceinline__ uint32_t tree_alloc_node(const GpuTreeDev& T) {
    int top = atomicSub(T.free_top, 1);
    if (top > 0) return T.free_list[top - 1];
    atomicAdd(T.free_top, 1);  // underflow repair
    uint32_t fresh = atomicAdd(T.next_node, 1u);
    if (fresh >= T.node_cap) {
        atomicMin(T.next_node, T.node_cap + 1024u);  // keep the counter bounded
        return NODE_SLOT_NONE;
    }
    return fresh;
}

__device__ __forceinline__ void tree_free_node(const GpuTreeDev& T, uint32_t id) {
    int top = atomicAdd(T.free_top, 1);
    if (top >= 0 && top < (int)T.node_cap) T.free_list[top] = id;
}

__device__ __forceinline__ unsigned long long mix64(unsigned long long x) {
    // splitmix64 finalizer
    x += 0x9E3779B97F4A7C15ull;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
    return x ^ (x >> 31);
}

// ---- chain-hash key schedule (host_tree.cpp computes the identical one) ----
// page_hash(page) = mix(sum_j mix(tok_j + SALT) * C^j  ^  page_size)
// chain(p)       = sum_{q<=p} page_hash(q) * W^q        (mod 2^64)
// key(p)         = max(2, mix(chain(p) ^ (p+1)*GOLD))
#define CHAIN_W 0xA24BAED4963EE407ull
#define CHAIN_GOLD 0x9E3779B97F4A7C15ull
#define CHAIN_SALT 0x5851F42D4C957F2Dull
#define MAX_LDS_PAGES 1024

__constant__ unsigned long long c_pow[WAVE];   // C^j, j < 64 (within-page)
__constant__ unsigned long long c_wpow[WAVE];  // W^p, p < 64 (within-chunk)
__constant__ unsigned long long c_w64[1];      // W^64 (chunk carry scale)

__device__ __forceinline__ unsigned long long lane_page_hash(const uint32_t* toks, int n) {
    unsigned long long h = 0;
    for (int j = 0; j < n; ++j)
        h += mix64((unsigned long long)toks[j] + CHAIN_SALT) * c_pow[j];
    return mix64(h ^ (unsigned long long)n);
}

__device__ __forceinline__ unsigned long long chain_key(unsigned long long chain, uint32_t depth) {
    unsigned long long k = mix64(chain ^ ((unsigned long long)(depth + 1) * CHAIN_GOLD));
    return k < 2 ? k + 2 : k;
}

__device__ __forceinline__ unsigned long long wave_prefix_sum_u64(unsigned long long v, int lane) {
#pragma unroll
    for (int off = 1; off < WAVE; off <<= 1) {
        unsigned long long up = __shfl_up(v, off, WAVE);
        if (lane >= off) v += up;
    }
    return v;
}

__device__ __forceinline__ unsigned long long atomic_load_key(const unsigned long long* p) {
    return __hip_atomic_load(p, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
}

// Per-lane linear probe: entry id if `key` is present, -1 at first empty.
__device__ int probe_find_lane(const GpuTreeDev& T, unsigned long long key) {
    uint32_t base = (uint32_t)(key & T.table_mask);
    for (uint32_t i = 0; i < 256u; ++i) {
        uint32_t slot = (base + i) & T.table_mask;
        unsigned long long k = atomic_load_key(&T.table_keys[slot]);
        if (k == key)
            return (int)__hip_atomic_load(&T.table_vals[slot], __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        if (k == EMPTY_KEY) return -1;
    }
    return -1;
}

// Insert `key`->`val` (lane-0 driven).  Two-step publication so a losing
// wave can never clobber the winner's val: CAS the key slot EMPTY->BUSY,
// store val, then release-store the real key.  A racing same-key insert that
// sees BUSY simply probes on and may create an unreachable duplicate node —
// bounded memory noise, never a wrong lookup.  Returns the node id placed in
// the slot (ours, or an existing same-key entry's), -1 when the table
// neighborhood is full.
#define BUSY_KEY 2ull
__device__ __forceinline__ bool try_claim(const GpuTreeDev& T, uint32_t slot,
                                          unsigned long long expected,
                                          unsigned long long key, uint32_t val) {
    if (__hip_atomic_compare_exchange_strong(&T.table_keys[slot], &expected, BUSY_KEY,
                                             __ATOMIC_ACQ_REL, __ATOMIC_RELAXED,
                                             __HIP_MEMORY_SCOPE_AGENT)) {
        __hip_atomic_store(&T.table_vals[slot], val, __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        __hip_atomic_store(&T.table_keys[slot], key, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_AGENT);
        return true;
    }
    return false;
}

__device__ int probe_insert(const GpuTreeDev& T, unsigned long long key, uint32_t val, uint32_t* out_slot) {
    uint32_t base = (uint32_t)(key & T.table_mask);
    // Pass 1: find the key (or the first EMPTY), remembering the first
    // tombstone.  A tombstone must NOT be claimed before the scan proves the
    // key is absent further down the chain — claiming it eagerly would
    // create a duplicate entry that shadows the live one (lookups stop at
    // the first match, so the live entry's tenant state would go dark).
    int first_ts = -1;
    uint32_t i = 0;
    for (; i < 1024u; ++i) {
        uint32_t slot = (base + i) & T.table_mask;
        // relaxed scan: acquire loads cost 2-3x per hop and invalidate the
        // CU's L1 chip-wide (MI355X_MICROARCH.md §polling); ordering comes
        // from the CAS in try_claim and the val-before-key publication
        unsigned long long k = atomic_load_key(&T.table_keys[slot]);
        if (k == key) {
            *out_slot = slot;
            return (int)__hip_atomic_load(&T.table_vals[slot], __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        }
        if (k == EMPTY_KEY) break;
        if (k == TOMBSTONE_KEY && first_ts < 0) first_ts = (int)slot;
        // BUSY or other key: probe on
    }
    // Pass 2: key absent — reuse the remembered tombstone first
    if (first_ts >= 0 && try_claim(T, (uint32_t)first_ts, TOMBSTONE_KEY, key, val)) {
        *out_slot = (uint32_t)first_ts;
        return (int)val;
    }
    // then fight for empty/tombstone slots from the scan frontier on
    for (; i < 1024u; ++i) {
        uint32_t slot = (base + i) & T.table_mask;
        unsigned long long k = atomic_load_key(&T.table_keys[slot]);
        if (k == key) {
            *out_slot = slot;
            return (int)__hip_atomic_load(&T.table_vals[slot], __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        }
        if ((k == EMPTY_KEY || k == TOMBSTONE_KEY) && try_claim(T, slot, k, key, val)) {
            *out_slot = slot;
            return (int)val;
        }
    }
    return -1;
}

// ---------------------------------------------------------------------------
// match + decide + insert, one wave per request
// ---------------------------------------------------------------------------
struct BatchArgs {
    const uint32_t* tokens;   // flattened
    const uint32_t* offsets;  // n_reqs + 1
    int n_reqs;
    // decision inputs
    unsigned long long healthy_mask;  // candidate worker slots
    int* loads;                       // [64] live load counters (device, atomically bumped)
    const int* processed;             // [64] processed_requests tie-break
    float cache_threshold;
    int imbalanced;  // batch-level trigger: force min-load
    int n_workers;
    int do_insert;
    int forced_tenant;  // >=0: skip decision, insert for this slot (insert-only)
    // mode 0: match+decide+insert (cache_aware).  mode 1: per-worker matched
    // depths only (KV-event overlap scoring, event_tree.rs:571 find_matches).
    // mode 2: clear forced_tenant's bit along the path (apply_removed).
    int mode;
    uint32_t* out_depths;  // mode 1: [n_reqs * 64] matched tokens per worker
    // outputs per request
    int* out_selected;       // worker slot (or -1)
    uint32_t* out_matched;   // matched token count
    uint32_t* out_tenant;    // matched (pre-decision) tenant slot or 0xffffffff
};

extern "C" __global__ void __launch_bounds__(WAVE)
smg_tree_match_insert(GpuTreeDev T, BatchArgs A) {
    int req = blockIdx.x;
    if (req >= A.n_reqs) return;
    int lane = threadIdx.x;

    uint32_t beg = A.offsets[req], end = A.offsets[req + 1];
    uint32_t n_tokens = end - beg;
    uint32_t n_pages = n_tokens / T.page_size;
    if (n_pages > T.max_pages) n_pages = T.max_pages;
    if (n_pages > MAX_LDS_PAGES) n_pages = MAX_LDS_PAGES;

    __shared__ unsigned long long s_keys[MAX_LDS_PAGES];
    __shared__ int s_ids[MAX_LDS_PAGES];

    // ---- phase A: parallel prefix match -----------------------------------
    // Lane p computes the chain key for depth (chunk*64 + p) and probes it;
    // all depths in a chunk resolve in ONE memory round trip instead of a
    // dependent walk.  matched = trailing-ones of the presence ballot.
    uint32_t matched_pages = 0;
    int deepest_id = -1;
    unsigned long long carry = 0;       // chain sum over previous chunks
    unsigned long long chunk_scale = 1; // W^(64*chunk)
    bool open = true;                   // no gap seen yet
    // mode 1 state: this lane owns worker slot `lane`
    uint32_t my_worker_pages = 0;
    bool my_worker_open = true;
    for (uint32_t base = 0; base < n_pages; base += WAVE) {
        uint32_t page = base + (uint32_t)lane;
        bool valid = page < n_pages;
        unsigned long long contrib = 0;
        if (valid)
            contrib = lane_page_hash(A.tokens + beg + page * T.page_size, (int)T.page_size) *
                      c_wpow[lane] * chunk_scale;
        unsigned long long scan = wave_prefix_sum_u64(contrib, lane);
        unsigned long long chain = carry + scan;
        unsigned long long key = valid ? chain_key(chain, page) : 0;
        int id = -1;
        unsigned long long tmask = 0;
        if (valid) {
            id = probe_find_lane(T, key);
            if (id >= 0)
                tmask = __hip_atomic_load(&T.node_tenants[id], __ATOMIC_RELAXED,
                                          __HIP_MEMORY_SCOPE_AGENT);
            s_keys[page] = key;
            s_ids[page] = id;
        }
        bool tenanted = tmask != 0ull;
        unsigned long long ok = __ballot(valid && id >= 0 && tenanted);
        uint32_t chunk_n = min(n_pages - base, (uint32_t)WAVE);
        uint32_t run = (~ok == 0ull) ? 64u : (uint32_t)(__ffsll((long long)~ok) - 1);
        if (run > chunk_n) run = chunk_n;
        if (open) {
            matched_pages += run;
            if (run > 0) {
                int last_id = __shfl(id, (int)(run - 1), WAVE);
                deepest_id = last_id;
            }
            if (run < chunk_n) open = false;
        }
        if (A.mode == 1) {
            // per-worker runs: 64 ballots transpose the lane-held tenant masks
            // into per-worker presence vectors; lane w folds worker w's run
            for (int w = 0; w < WAVE; ++w) {
                unsigned long long okw = __ballot(valid && id >= 0 && ((tmask >> w) & 1ull));
                if (lane == w && my_worker_open) {
                    uint32_t runw = (~okw == 0ull) ? 64u : (uint32_t)(__ffsll((long long)~okw) - 1);
                    if (runw > chunk_n) runw = chunk_n;
                    my_worker_pages += runw;
                    if (runw < chunk_n) my_worker_open = false;
                }
            }
        }
        // chunk carry: total of this chunk's contributions (lane 63's scan)
        carry += __shfl(scan, WAVE - 1, WAVE);
        chunk_scale *= c_w64[0];
    }
    __syncthreads();
    if (A.mode == 1) {
        A.out_depths[(size_t)req * WAVE + lane] = my_worker_pages * T.page_size;
        if (lane == 0) {
            A.out_matched[req] = matched_pages * T.page_size;
            A.out_selected[req] = -1;
            A.out_tenant[req] = 0xffffffffu;
        }
        return;
    }
    if (A.mode == 2) {
        // clear forced_tenant's attribution along the matched path
        unsigned long long clear = ~(1ull << A.forced_tenant);
        for (uint32_t page = lane; page < n_pages; page += WAVE) {
            int id = s_ids[page];
            if (id >= 0) {
                atomicAnd(&T.node_tenants[id], clear);
                T.node_ts[(size_t)id * WAVE + A.forced_tenant] = 0;
            }
        }
        if (lane == 0) {
            A.out_selected[req] = -1;
            A.out_matched[req] = matched_pages * T.page_size;
            A.out_tenant[req] = 0xffffffffu;
        }
        return;
    }
    uint32_t matched_tokens = matched_pages * T.page_size;
    uint32_t deepest_tenanted = deepest_id >= 0 ? (uint32_t)deepest_id : 0xffffffffu;
    unsigned long long deepest_mask =
        deepest_id >= 0
            ? __hip_atomic_load(&T.node_tenants[deepest_id], __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT)
            : 0ull;

    // MRU healthy tenant of the deepest matched node (lane-parallel argmax ts)
    int matched_tenant = -1;
    if (deepest_tenanted != 0xffffffffu) {
        unsigned long long m = deepest_mask & A.healthy_mask;
        uint32_t my_ts = 0;
        int my_slot = -1;
        if (lane < 64 && ((m >> lane) & 1ull)) {
            my_ts = __hip_atomic_load(&T.node_ts[(size_t)deepest_tenanted * WAVE + lane],
                                      __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
            my_slot = lane;
        }
        // wave argmax (ts, slot)
        unsigned long long packed = ((unsigned long long)my_ts << 32) | (unsigned)(my_slot + 1);
        if (my_slot < 0) packed = 0;
#pragma unroll
        for (int off = 32; off > 0; off >>= 1) {
            unsigned long long other = __shfl_xor(packed, off, WAVE);
            if (other > packed) packed = other;
        }
        if ((packed & 0xffffffffu) != 0) matched_tenant = (int)(packed & 0xffffffffu) - 1;
        // touch the matched tenant's stamp along the WHOLE matched path
        // (mirrors token_tree.rs:693 touch_tenant on match): refreshing only
        // the deepest node would let the eviction sweep clear the shallow
        // nodes of a hot path out from under it, breaking its match run
        if (matched_tenant >= 0) {
            uint32_t now = 0;
            if (lane == 0) now = atomicAdd(T.clock_, 1u) + 1u;
            now = __shfl(now, 0, WAVE);
            for (uint32_t page = lane; page < matched_pages; page += WAVE) {
                int pid = s_ids[page];
                if (pid >= 0)
                    atomicMax(&T.node_ts[(size_t)pid * WAVE + matched_tenant], now);
            }
        }
    }

    // ---- decision (cache_aware.rs:986 semantics) ---------------------------
    int selected;
    if (A.forced_tenant >= 0) {
        selected = A.forced_tenant;
    } else {
        float rate = n_tokens ? (float)matched_tokens / (float)n_tokens : 0.f;
        bool hit = !A.imbalanced && rate > A.cache_threshold && matched_tenant >= 0;
        if (hit) {
            selected = matched_tenant;
        } else {
            // min-load over healthy slots, tie-break (load, processed, slot)
            int my_slot = -1;
            long long my_key = 0x7fffffffffffffffll;
            if (lane < A.n_workers && ((A.healthy_mask >> lane) & 1ull)) {
                int ld = __hip_atomic_load(&A.loads[lane], __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
                int pr = A.processed[lane];
                my_key = ((long long)ld << 40) | ((long long)(pr & 0xffffff) << 16) | lane;
                my_slot = lane;
            }
#pragma unroll
            for (int off = 32; off > 0; off >>= 1) {
                long long ok = __shfl_xor(my_key, off, WAVE);
                int os = __shfl_xor(my_slot, off, WAVE);
                if (ok < my_key) { my_key = ok; my_slot = os; }
            }
            selected = my_slot;
        }
        if (lane == 0 && selected >= 0)
            atomicAdd(&A.loads[selected], 1);  // intra-batch load guard
    }

    if (lane == 0) {
        A.out_selected[req] = selected;
        A.out_matched[req] = matched_tokens;
        A.out_tenant[req] = (matched_tenant >= 0) ? (uint32_t)matched_tenant : 0xffffffffu;
    }
    if (!A.do_insert || selected < 0) return;

    // ---- phase B: parallel insert of the whole path for `selected` ---------
    // Each lane owns one page depth: reuse the id probed in phase A, or
    // allocate+publish a fresh entry, then attribute tenant bit + LRU stamp.
    // Fully parallel — entries are independent under the chain-key scheme.
    uint32_t now = 0;
    if (lane == 0) now = atomicAdd(T.clock_, 1u) + 1u;
    now = __shfl(now, 0, WAVE);
    unsigned long long tbit = 1ull << selected;
    for (uint32_t base = 0; base < n_pages; base += WAVE) {
        uint32_t page = base + (uint32_t)lane;
        if (page >= n_pages) continue;
        int id = s_ids[page];
        if (id < 0) {
            uint32_t fresh = tree_alloc_node(T);
            if (fresh == NODE_SLOT_NONE) continue;  // pool exhausted until next reclaim
            uint32_t oslot;
            id = probe_insert(T, s_keys[page], fresh, &oslot);
            if (id < 0) {
                tree_free_node(T, fresh);  // table neighborhood full
                continue;
            }
            if (id == (int)fresh)
                T.node_slot[fresh] = oslot;  // we published it: remember the slot for reclaim
            else
                tree_free_node(T, fresh);  // same-key entry won the race; recycle ours
        }
        atomicOr(&T.node_tenants[id], tbit);
        atomicMax(&T.node_ts[(size_t)id * WAVE + selected], now);
    }
}

// ---------------------------------------------------------------------------
// maintenance kernels
// ---------------------------------------------------------------------------
extern "C" __global__ void smg_tree_remove_tenant(GpuTreeDev T, int slot_idx) {
    uint32_t n = min(*T.next_node, T.node_cap);
    unsigned long long mask = ~(1ull << slot_idx);
    for (uint32_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += gridDim.x * blockDim.x) {
        atomicAnd(&T.node_tenants[i], mask);
        T.node_ts[(size_t)i * WAVE + slot_idx] = 0;
    }
}

// Clear tenant bits whose stamp is older than `cutoff`; nodes left with no
// tenants block matches through them (eviction semantics of token_tree.rs).
extern "C" __global__ void smg_tree_evict_older(GpuTreeDev T, uint32_t cutoff) {
    uint32_t n = min(*T.next_node, T.node_cap);
    uint32_t node = blockIdx.x;  // one wave per node, lane per tenant slot
    int lane = threadIdx.x;
    for (uint32_t i = node; i < n; i += gridDim.x) {
        unsigned long long tenants =
            __hip_atomic_load(&T.node_tenants[i], __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        if (!tenants) continue;
        if (((tenants >> lane) & 1ull) && T.node_ts[(size_t)i * WAVE + lane] < cutoff) {
            atomicAnd(&T.node_tenants[i], ~(1ull << lane));
        }
    }
}

// Reclaim nodes whose tenant mask went to zero (evict sweep / tenant removal
// left them unreachable): tombstone the table slot, clear the LRU row, and
// push the node id onto the free list for reuse.  One wave per node; runs
// stream-ordered with match/insert so no extra synchronization is needed.
// Reference behavior: token_tree.rs eviction frees nodes back to the pool so
// the tree keeps learning at --max-tree-size.
extern "C" __global__ void __launch_bounds__(WAVE)
smg_tree_reclaim(GpuTreeDev T, unsigned long long* out_reclaimed) {
    uint32_t n = min(*T.next_node, T.node_cap);
    int lane = threadIdx.x;
    for (uint32_t i = blockIdx.x; i < n; i += gridDim.x) {
        if (i == 0) continue;  // root
        unsigned long long tenants =
            __hip_atomic_load(&T.node_tenants[i], __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        uint32_t slot = T.node_slot[i];
        if (tenants != 0ull || slot == NODE_SLOT_NONE) continue;
        // clear the per-tenant LRU row so a reused node starts cold
        T.node_ts[(size_t)i * WAVE + lane] = 0;
        if (lane == 0) {
            if (__hip_atomic_load(&T.table_vals[slot], __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT) == i)
                __hip_atomic_store(&T.table_keys[slot], TOMBSTONE_KEY, __ATOMIC_RELEASE,
                                   __HIP_MEMORY_SCOPE_AGENT);
            T.node_slot[i] = NODE_SLOT_NONE;
            tree_free_node(T, i);
            if (out_reclaimed) atomicAdd(out_reclaimed, 1ull);
        }
    }
}

// Clear one tenant's bit on explicit entries (KV-event apply_removed: the
// host resolves removed block hashes to their stored chain keys).
extern "C" __global__ void smg_tree_clear_entries(GpuTreeDev T, const unsigned long long* keys,
                                                  int n, int slot_idx) {
    int i = blockIdx.x * blockDim.x + threadIdx.x;
    if (i >= n) return;
    int id = probe_find_lane(T, keys[i]);
    if (id >= 0) {
        atomicAnd(&T.node_tenants[id], ~(1ull << slot_idx));
        T.node_ts[(size_t)id * WAVE + slot_idx] = 0;
    }
}

// Count live nodes / per-tenant attributed nodes (stats + tie-breaks).
extern "C" __global__ void smg_tree_stats(GpuTreeDev T, unsigned long long* out_counts /*65*/) {
    uint32_t n = min(*T.next_node, T.node_cap);
    int lane = threadIdx.x;
    unsigned long long live = 0, mine = 0;
    // grid-stride over nodes; one wave per node, lane = tenant slot
    for (uint32_t i = blockIdx.x; i < n; i += gridDim.x) {
        unsigned long long tenants =
            __hip_atomic_load(&T.node_tenants[i], __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
        if (lane == 0 && tenants) live++;
        if ((tenants >> lane) & 1ull) mine++;
    }
    if (lane == 0 && live) atomicAdd(&out_counts[64], live);
    if (mine) atomicAdd(&out_counts[lane], mine);
}

// ---------------------------------------------------------------------------
// host-side management (exported C API; pybind in bindings.cpp)
// ---------------------------------------------------------------------------
#define HIP_CHECK(x)                                                                   \
    do {                                                                               \
        hipError_t _e = (x);                                                           \
        if (_e != hipSuccess) {                                                        \
            fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(_e), __FILE__, __LINE__); \
            return nullptr;                                                            \
        }                                                                              \
    } while (0)

struct GpuTreeHost {
    GpuTreeDev dev;
    hipStream_t stream;
    uint32_t table_size;
    // pinned staging
    uint32_t* h_tokens;
    uint32_t* h_offsets;
    int* h_selected;
    uint32_t* h_matched;
    uint32_t* h_tenant;
    uint32_t max_batch_tokens;
    uint32_t max_batch_reqs;
    // device batch buffers
    uint32_t* d_tokens;
    uint32_t* d_offsets;
    int* d_selected;
    uint32_t* d_matched;
    uint32_t* d_tenant;
    int* d_loads;
    int* d_processed;
    unsigned long long* d_counts;
    uint32_t* d_depths;
};

extern "C" void* smg_gpu_tree_create(int device, uint32_t node_cap, uint32_t table_size,
                                     uint32_t page_size, uint32_t max_pages,
                                     uint32_t max_batch_reqs, uint32_t max_batch_tokens) {
    if (hipSetDevice(device) != hipSuccess) return nullptr;
    GpuTreeHost* t = new GpuTreeHost{};
    t->table_size = table_size;
    t->max_batch_reqs = max_batch_reqs;
    t->max_batch_tokens = max_batch_tokens;
    GpuTreeDev& D = t->dev;
    D.node_cap = node_cap;
    D.table_mask = table_size - 1;
    D.page_size = page_size;
    D.max_pages = max_pages;
    HIP_CHECK(hipStreamCreate(&t->stream));
    HIP_CHECK(hipMalloc(&D.table_keys, sizeof(unsigned long long) * table_size));
    HIP_CHECK(hipMalloc(&D.table_vals, sizeof(uint32_t) * table_size));
    HIP_CHECK(hipMalloc(&D.node_tenants, sizeof(unsigned long long) * node_cap));
    HIP_CHECK(hipMalloc(&D.node_ts, sizeof(uint32_t) * (size_t)node_cap * WAVE));
    HIP_CHECK(hipMalloc(&D.node_parent, sizeof(uint32_t) * node_cap));
    HIP_CHECK(hipMalloc(&D.node_slot, sizeof(uint32_t) * node_cap));
    HIP_CHECK(hipMalloc(&D.free_list, sizeof(uint32_t) * node_cap));
    HIP_CHECK(hipMalloc(&D.free_top, sizeof(int)));
    HIP_CHECK(hipMalloc(&D.next_node, sizeof(uint32_t) * 2));
    D.clock_ = D.next_node + 1;
    HIP_CHECK(hipMemset(D.table_keys, 0, sizeof(unsigned long long) * table_size));
    HIP_CHECK(hipMemset(D.node_tenants, 0, sizeof(unsigned long long) * node_cap));
    HIP_CHECK(hipMemset(D.node_ts, 0, sizeof(uint32_t) * (size_t)node_cap * WAVE));
    HIP_CHECK(hipMemset(D.node_slot, 0xff, sizeof(uint32_t) * node_cap));  // NODE_SLOT_NONE
    HIP_CHECK(hipMemset(D.free_top, 0, sizeof(int)));
    uint32_t init[2] = {1u, 0u};  // node 0 = root
    HIP_CHECK(hipMemcpy(D.next_node, init, sizeof(init), hipMemcpyHostToDevice));
    // polynomial powers: C^j for the in-page hash, W^p + W^64 for the chain
    unsigned long long pows[WAVE], wpows[WAVE], w64;
    pows[0] = 1ull;
    for (int i = 1; i < WAVE; ++i) pows[i] = pows[i - 1] * 0x100000001B3ull;
    wpows[0] = 1ull;
    for (int i = 1; i < WAVE; ++i) wpows[i] = wpows[i - 1] * CHAIN_W;
    w64 = wpows[WAVE - 1] * CHAIN_W;
    HIP_CHECK(hipMemcpyToSymbol(HIP_SYMBOL(c_pow), pows, sizeof(pows)));
    HIP_CHECK(hipMemcpyToSymbol(HIP_SYMBOL(c_wpow), wpows, sizeof(wpows)));
    HIP_CHECK(hipMemcpyToSymbol(HIP_SYMBOL(c_w64), &w64, sizeof(w64)));
    // batch buffers
    HIP_CHECK(hipHostMalloc((void**)&t->h_tokens, sizeof(uint32_t) * max_batch_tokens));
    HIP_CHECK(hipHostMalloc((void**)&t->h_offsets, sizeof(uint32_t) * (max_batch_reqs + 1)));
    HIP_CHECK(hipHostMalloc((void**)&t->h_selected, sizeof(int) * max_batch_reqs));
    HIP_CHECK(hipHostMalloc((void**)&t->h_matched, sizeof(uint32_t) * max_batch_reqs));
    HIP_CHECK(hipHostMalloc((void**)&t->h_tenant, sizeof(uint32_t) * max_batch_reqs));
    HIP_CHECK(hipMalloc(&t->d_tokens, sizeof(uint32_t) * max_batch_tokens));
    HIP_CHECK(hipMalloc(&t->d_offsets, sizeof(uint32_t) * (max_batch_reqs + 1)));
    HIP_CHECK(hipMalloc(&t->d_selected, sizeof(int) * max_batch_reqs));
    HIP_CHECK(hipMalloc(&t->d_matched, sizeof(uint32_t) * max_batch_reqs));
    HIP_CHECK(hipMalloc(&t->d_tenant, sizeof(uint32_t) * max_batch_reqs));
    HIP_CHECK(hipMalloc(&t->d_loads, sizeof(int) * WAVE));
    HIP_CHECK(hipMalloc(&t->d_processed, sizeof(int) * WAVE));
    HIP_CHECK(hipMalloc(&t->d_counts, sizeof(unsigned long long) * 65));
    HIP_CHECK(hipMalloc(&t->d_depths, sizeof(uint32_t) * (size_t)max_batch_reqs * 64));
    return t;
}

extern "C" void smg_gpu_tree_destroy(void* p) {
    if (!p) return;
    GpuTreeHost* t = (GpuTreeHost*)p;
    hipFree(t->dev.table_keys); hipFree(t->dev.table_vals);
    hipFree(t->dev.node_tenants); hipFree(t->dev.node_ts);
    hipFree(t->dev.node_parent); hipFree(t->dev.node_slot);
    hipFree(t->dev.free_list); hipFree(t->dev.free_top);
    hipFree(t->dev.next_node);
    hipHostFree(t->h_tokens); hipHostFree(t->h_offsets); hipHostFree(t->h_selected);
    hipHostFree(t->h_matched); hipHostFree(t->h_tenant);
    hipFree(t->d_tokens); hipFree(t->d_offsets); hipFree(t->d_selected);
    hipFree(t->d_matched); hipFree(t->d_tenant); hipFree(t->d_loads);
    hipFree(t->d_processed); hipFree(t->d_counts); hipFree(t->d_depths);
    hipStreamDestroy(t->stream);
    delete t;
}

// Batched match/decide/insert.  tokens/offsets are caller-filled into the
// pinned buffers via smg_gpu_tree_staging().  Returns 0 on success.
extern "C" int smg_gpu_tree_run(void* p, int n_reqs, unsigned long long healthy_mask,
                                const int* loads, const int* processed, int n_workers,
                                float cache_threshold, int imbalanced, int do_insert,
                                int forced_tenant, int mode,
                                int* out_selected, uint32_t* out_matched, uint32_t* out_tenant,
                                uint32_t* out_depths) {
    GpuTreeHost* t = (GpuTreeHost*)p;
    if (n_reqs <= 0 || (uint32_t)n_reqs > t->max_batch_reqs) return -1;
    uint32_t n_tokens = t->h_offsets[n_reqs];
    if (n_tokens > t->max_batch_tokens) return -2;
    hipStream_t s = t->stream;
    hipMemcpyAsync(t->d_tokens, t->h_tokens, sizeof(uint32_t) * n_tokens, hipMemcpyHostToDevice, s);
    hipMemcpyAsync(t->d_offsets, t->h_offsets, sizeof(uint32_t) * (n_reqs + 1), hipMemcpyHostToDevice, s);
    hipMemcpyAsync(t->d_loads, loads, sizeof(int) * WAVE, hipMemcpyHostToDevice, s);
    hipMemcpyAsync(t->d_processed, processed, sizeof(int) * WAVE, hipMemcpyHostToDevice, s);
    BatchArgs A{};
    A.tokens = t->d_tokens;
    A.offsets = t->d_offsets;
    A.n_reqs = n_reqs;
    A.healthy_mask = healthy_mask;
    A.loads = t->d_loads;
    A.processed = t->d_processed;
    A.cache_threshold = cache_threshold;
    A.imbalanced = imbalanced;
    A.n_workers = n_workers;
    A.do_insert = do_insert;
    A.forced_tenant = forced_tenant;
    A.mode = mode;
    A.out_depths = t->d_depths;
    A.out_selected = t->d_selected;
    A.out_matched = t->d_matched;
    A.out_tenant = t->d_tenant;
    hipLaunchKernelGGL(smg_tree_match_insert, dim3(n_reqs), dim3(WAVE), 0, s, t->dev, A);
    hipMemcpyAsync(t->h_selected, t->d_selected, sizeof(int) * n_reqs, hipMemcpyDeviceToHost, s);
    hipMemcpyAsync(t->h_matched, t->d_matched, sizeof(uint32_t) * n_reqs, hipMemcpyDeviceToHost, s);
    hipMemcpyAsync(t->h_tenant, t->d_tenant, sizeof(uint32_t) * n_reqs, hipMemcpyDeviceToHost, s);
    if (mode == 1 && out_depths)
        hipMemcpyAsync(out_depths, t->d_depths, sizeof(uint32_t) * (size_t)n_reqs * 64,
                       hipMemcpyDeviceToHost, s);
    if (hipStreamSynchronize(s) != hipSuccess) return -3;
    for (int i = 0; i < n_reqs; ++i) {
        out_selected[i] = t->h_selected[i];
        out_matched[i] = t->h_matched[i];
        out_tenant[i] = t->h_tenant[i];
    }
    return 0;
}

extern "C" void smg_gpu_tree_staging(void* p, uint32_t** tokens, uint32_t** offsets) {
    GpuTreeHost* t = (GpuTreeHost*)p;
    *tokens = t->h_tokens;
    *offsets = t->h_offsets;
}

extern "C" int smg_gpu_tree_clear_entries(void* p, const unsigned long long* keys, int n, int slot) {
    GpuTreeHost* t = (GpuTreeHost*)p;
    if (n <= 0) return 0;
    unsigned long long* d_keys = nullptr;
    if (hipMalloc(&d_keys, sizeof(unsigned long long) * n) != hipSuccess) return -1;
    hipMemcpyAsync(d_keys, keys, sizeof(unsigned long long) * n, hipMemcpyHostToDevice, t->stream);
    hipLaunchKernelGGL(smg_tree_clear_entries, dim3((n + 255) / 256), dim3(256), 0, t->stream,
                       t->dev, d_keys, n, slot);
    int rc = hipStreamSynchronize(t->stream) == hipSuccess ? 0 : -2;
    hipFree(d_keys);
    return rc;
}

extern "C" int smg_gpu_tree_remove_tenant(void* p, int slot) {
    GpuTreeHost* t = (GpuTreeHost*)p;
    hipLaunchKernelGGL(smg_tree_remove_tenant, dim3(256), dim3(256), 0, t->stream, t->dev, slot);
    return hipStreamSynchronize(t->stream) == hipSuccess ? 0 : -1;
}

extern "C" int smg_gpu_tree_evict_older(void* p, uint32_t cutoff) {
    GpuTreeHost* t = (GpuTreeHost*)p;
    hipLaunchKernelGGL(smg_tree_evict_older, dim3(2048), dim3(WAVE), 0, t->stream, t->dev, cutoff);
    return hipStreamSynchronize(t->stream) == hipSuccess ? 0 : -1;
}

// Sweep tenant-less nodes back onto the free list; returns the count (or -1).
extern "C" long long smg_gpu_tree_reclaim(void* p) {
    GpuTreeHost* t = (GpuTreeHost*)p;
    hipMemsetAsync(t->d_counts, 0, sizeof(unsigned long long), t->stream);
    hipLaunchKernelGGL(smg_tree_reclaim, dim3(2048), dim3(WAVE), 0, t->stream, t->dev, t->d_counts);
    unsigned long long n = 0;
    hipMemcpyAsync(&n, t->d_counts, sizeof(n), hipMemcpyDeviceToHost, t->stream);
    if (hipStreamSynchronize(t->stream) != hipSuccess) return -1;
    return (long long)n;
}

// out[0..63] per-tenant node counts, out[64] live nodes, out[65] allocated,
// out[66] clock, out[67] free-list depth
extern "C" int smg_gpu_tree_stats(void* p, unsigned long long* out /*68*/) {
    GpuTreeHost* t = (GpuTreeHost*)p;
    hipMemsetAsync(t->d_counts, 0, sizeof(unsigned long long) * 65, t->stream);
    hipLaunchKernelGGL(smg_tree_stats, dim3(1024), dim3(WAVE), 0, t->stream, t->dev, t->d_counts);
    hipMemcpyAsync(out, t->d_counts, sizeof(unsigned long long) * 65, hipMemcpyDeviceToHost, t->stream);
    uint32_t counters[2];
    int free_top = 0;
    hipMemcpyAsync(counters, t->dev.next_node, sizeof(counters), hipMemcpyDeviceToHost, t->stream);
    hipMemcpyAsync(&free_top, t->dev.free_top, sizeof(int), hipMemcpyDeviceToHost, t->stream);
    if (hipStreamSynchronize(t->stream) != hipSuccess) return -1;
    out[65] = counters[0];
    out[66] = counters[1];
    out[67] = (unsigned long long)(free_top > 0 ? free_top : 0);
    return 0;
}

extern "C" int smg_gpu_tree_clear(void* p) {
    GpuTreeHost* t = (GpuTreeHost*)p;
    hipMemsetAsync(t->dev.table_keys, 0, sizeof(unsigned long long) * t->table_size, t->stream);
    hipMemsetAsync(t->dev.node_tenants, 0, sizeof(unsigned long long) * t->dev.node_cap, t->stream);
    hipMemsetAsync(t->dev.node_ts, 0, sizeof(uint32_t) * (size_t)t->dev.node_cap * WAVE, t->stream);
    hipMemsetAsync(t->dev.node_slot, 0xff, sizeof(uint32_t) * t->dev.node_cap, t->stream);
    hipMemsetAsync(t->dev.free_top, 0, sizeof(int), t->stream);
    uint32_t init[2] = {1u, 0u};
    hipMemcpyAsync(t->dev.next_node, init, sizeof(init), hipMemcpyHostToDevice, t->stream);
    return hipStreamSynchronize(t->stream) == hipSuccess ? 0 : -1;
}

extern "C" int smg_hip_device_count() {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n;
}
