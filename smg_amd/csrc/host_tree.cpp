// Host-side paged radix prefix tree — the fast CPU twin of gpu_tree.hip.
//
// Identical semantics to smg_amd/kvindex/pytree.py (itself modeled on the
// reference's crates/kv_index/src/token_tree.rs:303): one page per node,
// children in one hash map keyed by (parent, page), per-(node,tenant) LRU
// stamps, page-aligned matching, pre-insert match resolution.  Used when no
// GPU is present and as the differential oracle for the device tree.
#include <algorithm>
#include <cstdint>
#include <cstring>
#include <queue>
#include <string>
#include <unordered_map>
#include <vector>

namespace smg {

static inline uint64_t mix64(uint64_t x) {
    x += 0x9E3779B97F4A7C15ull;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
    return x ^ (x >> 31);
}

// chain-key schedule — MUST match gpu_tree.hip exactly
static const uint64_t CHAIN_W = 0xA24BAED4963EE407ull;
static const uint64_t CHAIN_GOLD = 0x9E3779B97F4A7C15ull;

static inline uint64_t chain_key_of(uint64_t chain, uint32_t depth) {
    uint64_t k = mix64(chain ^ ((uint64_t)(depth + 1) * CHAIN_GOLD));
    return k < 2 ? k + 2 : k;
}

struct MatchOut {
    int tenant = -1;  // tenant slot, -1 = none
    uint32_t matched = 0;
    uint32_t input = 0;
};

class HostTree {
   public:
    explicit HostTree(uint32_t page_size) : page_size_(page_size) {
        nodes_.push_back(Node{});  // root, id 0
    }

    // page hash identical to the device kernel (order-sensitive polynomial)
    uint64_t page_hash(const uint32_t* toks, uint32_t n) const {
        uint64_t h = 0;
        uint64_t pw = 1;
        for (uint32_t i = 0; i < n; ++i) {
            h += mix64((uint64_t)toks[i] + 0x5851F42D4C957F2Dull) * pw;
            pw *= 0x100000001B3ull;
        }
        return mix64(h ^ (uint64_t)n);
    }

    MatchOut match(const uint32_t* toks, uint32_t n, bool touch = true) {
        MatchOut out;
        out.input = n;
        uint32_t pages = n / page_size_;
        uint32_t now = ++clock_;
        uint64_t chain = 0, wpow = 1;
        for (uint32_t p = 0; p < pages; ++p) {
            chain += page_hash(toks + p * page_size_, page_size_) * wpow;
            wpow *= CHAIN_W;
            uint64_t key = chain_key_of(chain, p);
            auto it = children_.find(key);
            if (it == children_.end()) break;
            Node& node = nodes_[it->second];
            if (node.tenant_mask == 0) break;
            // MRU tenant
            int best = -1;
            uint32_t best_ts = 0;
            for (int t = 0; t < 64; ++t) {
                if ((node.tenant_mask >> t) & 1ull) {
                    uint32_t ts = node.ts[t];
                    if (best < 0 || ts > best_ts || (ts == best_ts && t > best)) {
                        best = t;
                        best_ts = ts;
                    }
                }
            }
            if (touch && best >= 0) node.ts[best] = now;
            out.tenant = best;
            out.matched = (p + 1) * page_size_;
        }
        return out;
    }

    // returns tokens newly attributed to `tenant`
    uint32_t insert(const uint32_t* toks, uint32_t n, int tenant) {
        uint32_t pages = n / page_size_;
        uint32_t cur = 0;
        uint32_t added = 0;
        uint32_t now = ++clock_;
        uint64_t bit = 1ull << tenant;
        uint64_t chain = 0, wpow = 1;
        for (uint32_t p = 0; p < pages; ++p) {
            chain += page_hash(toks + p * page_size_, page_size_) * wpow;
            wpow *= CHAIN_W;
            uint64_t key = chain_key_of(chain, p);
            auto it = children_.find(key);
            uint32_t nid;
            if (it == children_.end()) {
                nid = (uint32_t)nodes_.size();
                Node node;
                node.parent = cur;
                nodes_.push_back(node);
                children_.emplace(key, nid);
                keys_.push_back(key);
                nodes_[cur].child_count++;
            } else {
                nid = it->second;
            }
            Node& node = nodes_[nid];
            if (!(node.tenant_mask & bit)) {
                added += page_size_;
                node.tenant_mask |= bit;
            }
            node.ts[tenant] = now;
            cur = nid;
        }
        tenant_tokens_[tenant] += added;
        return added;
    }

    void remove_tenant(int tenant) {
        uint64_t mask = ~(1ull << tenant);
        for (auto& n : nodes_) {
            n.tenant_mask &= mask;
            n.ts[tenant] = 0;
        }
        tenant_tokens_[tenant] = 0;
    }

    // LRU-evict (leaf, tenant) attributions until <= max_nodes live nodes.
    uint32_t evict(size_t max_nodes) {
        uint32_t removed = 0;
        if (live_count() <= max_nodes) return 0;
        // (ts, node, tenant) min-heap over leaf attributions
        using Item = std::tuple<uint32_t, uint32_t, int>;
        std::priority_queue<Item, std::vector<Item>, std::greater<Item>> heap;
        for (uint32_t i = 1; i < nodes_.size(); ++i) {
            if (!nodes_[i].dead && nodes_[i].child_count == 0) {
                if (nodes_[i].tenant_mask == 0) {
                    heap.emplace(0u, i, -1);
                } else {
                    for (int t = 0; t < 64; ++t)
                        if ((nodes_[i].tenant_mask >> t) & 1ull) heap.emplace(nodes_[i].ts[t], i, t);
                }
            }
        }
        while (live_count() > max_nodes && !heap.empty()) {
            auto [ts, nid, t] = heap.top();
            heap.pop();
            Node& node = nodes_[nid];
            if (node.dead || node.child_count != 0) continue;
            if (t >= 0 && ((node.tenant_mask >> t) & 1ull)) {
                node.tenant_mask &= ~(1ull << t);
                if (tenant_tokens_[t] >= page_size_) tenant_tokens_[t] -= page_size_;
            }
            if (node.tenant_mask == 0) {
                node.dead = true;
                children_.erase(keys_[nid - 1]);
                removed++;
                dead_count_++;
                Node& parent = nodes_[node.parent];
                if (--parent.child_count == 0 && node.parent != 0 && !parent.dead) {
                    if (parent.tenant_mask == 0) {
                        heap.emplace(0u, node.parent, -1);
                    } else {
                        for (int pt = 0; pt < 64; ++pt)
                            if ((parent.tenant_mask >> pt) & 1ull)
                                heap.emplace(parent.ts[pt], node.parent, pt);
                    }
                }
            }
        }
        return removed;
    }

    void clear() {
        nodes_.clear();
        children_.clear();
        keys_.clear();
        tenant_tokens_.clear();
        dead_count_ = 0;
        clock_ = 0;
        nodes_.push_back(Node{});
    }

    size_t live_count() const { return nodes_.size() - 1 - dead_count_; }
    uint64_t tenant_tokens(int t) {
        auto it = tenant_tokens_.find(t);
        return it == tenant_tokens_.end() ? 0 : it->second;
    }
    uint32_t page_size() const { return page_size_; }

   private:
    struct Node {
        uint32_t parent = 0;
        uint32_t child_count = 0;
        uint64_t tenant_mask = 0;
        bool dead = false;
        uint32_t ts[64] = {0};
    };

    static uint64_t child_key(uint32_t parent, uint64_t ph) {
        uint64_t k = mix64(((uint64_t)parent << 32) ^ ph * 0x9E3779B97F4A7C15ull);
        return k < 2 ? k + 2 : k;
    }

    uint32_t page_size_;
    uint32_t clock_ = 0;
    size_t dead_count_ = 0;
    std::vector<Node> nodes_;
    std::vector<uint64_t> keys_;  // node id-1 -> its child key
    std::unordered_map<uint64_t, uint32_t> children_;
    std::unordered_map<int, uint64_t> tenant_tokens_;
};

}  // namespace smg
