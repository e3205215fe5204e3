// Sanitizer self-test for the host radix tree (SURVEY §5.2 discipline: the
// C++/HIP rebuild must add its own memory/UB checking since the language no
// longer proves safety).  Built by csrc/build.py build_sanitized() with
// -fsanitize=address,undefined and run in CPU CI (tests/test_sanitizer.py);
// any heap overflow / use-after-free / UB in HostTree aborts the test.
//
// The workload mirrors the differential tests: randomized shared-prefix
// inserts and matches across tenants, interleaved with eviction, tenant
// removal and clears — the full mutation surface of host_tree.cpp.
#include <cassert>
#include <cstdint>
#include <cstdio>
#include <random>
#include <vector>

#include "host_tree.cpp"

using smg::HostTree;
using smg::MatchOut;

int main() {
    std::mt19937 rng(42);
    for (int page_size : {4, 8, 16}) {
        HostTree tree((uint32_t)page_size);
        std::vector<std::vector<uint32_t>> prefixes;
        for (int i = 0; i < 8; ++i) {
            std::vector<uint32_t> p(page_size * (2 + (int)(rng() % 6)));
            for (auto& t : p) t = rng() % 1000;
            prefixes.push_back(p);
        }
        for (int iter = 0; iter < 4000; ++iter) {
            auto seq = prefixes[rng() % prefixes.size()];
            int tail = (int)(rng() % 40);
            for (int j = 0; j < tail; ++j) seq.push_back(rng() % 1000);
            int tenant = (int)(rng() % 64);
            MatchOut m = tree.match(seq.data(), (uint32_t)seq.size());
            assert(m.matched <= seq.size());
            assert(m.matched % page_size == 0);
            tree.insert(seq.data(), (uint32_t)seq.size(), tenant);
            // a just-inserted path must fully match
            MatchOut m2 = tree.match(seq.data(), (uint32_t)seq.size(), false);
            assert(m2.matched == (seq.size() / page_size) * page_size);
            if (iter % 257 == 0) tree.evict(64);
            if (iter % 611 == 0) tree.remove_tenant(tenant);
            if (iter % 1501 == 0) tree.clear();
        }
        // zero-length + single-token edge cases
        uint32_t one = 7;
        assert(tree.match(&one, 0).matched == 0);
        tree.insert(&one, 1, 0);  // below one page: no-op path
        assert(tree.match(&one, 1).matched == 0);
        // full eviction leaves a consistent tree
        tree.evict(0);
        std::vector<uint32_t> p = prefixes[0];
        tree.insert(p.data(), (uint32_t)p.size(), 3);
        assert(tree.match(p.data(), (uint32_t)p.size()).matched > 0);
    }
    printf("host_tree sanitizer self-test OK\n");
    return 0;
}
