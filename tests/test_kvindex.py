"""Paged radix tree semantics (reference: crates/kv_index token_tree.rs tests).
These same cases run against the C++/GPU backends differentially
(test_kvindex_gpu.py)."""
from smg_amd.kvindex.pytree import PagedRadixTree, StringTree


def tree(ps=4):
    return PagedRadixTree(page_size=ps)


class TestMatch:
    def test_empty_tree_no_match(self):
        t = tree()
        r = t.match(list(range(16)))
        assert r.matched_token_count == 0
        assert r.tenant is None

    def test_short_input_below_page(self):
        t = tree(ps=16)
        t.insert(list(range(32)), "w0")
        r = t.match(list(range(8)))  # < one page
        assert r.matched_token_count == 0

    def test_exact_match(self):
        t = tree()
        toks = list(range(16))
        t.insert(toks, "w0")
        r = t.match(toks)
        assert r.matched_token_count == 16
        assert r.tenant == "w0"
        assert r.match_rate == 1.0

    def test_page_aligned_truncation(self):
        t = tree()
        t.insert(list(range(16)), "w0")
        r = t.match(list(range(18)))  # 18 -> 16 aligned
        assert r.matched_token_count == 16
        assert r.input_token_count == 18

    def test_partial_prefix(self):
        t = tree()
        t.insert(list(range(16)), "w0")
        q = list(range(8)) + [99, 98, 97, 96] + list(range(12, 16))
        r = t.match(q)
        assert r.matched_token_count == 8

    def test_deepest_tenant_wins(self):
        t = tree()
        toks = list(range(16))
        t.insert(toks[:8], "w0")
        t.insert(toks, "w1")
        r = t.match(toks)
        assert r.matched_token_count == 16
        assert r.tenant == "w1"

    def test_mru_tenant_preferred_on_tie(self):
        t = tree()
        toks = list(range(8))
        t.insert(toks, "w0")
        t.insert(toks, "w1")  # w1 touched later
        assert t.match(toks).tenant == "w1"


class TestInsert:
    def test_added_counts_new_tokens_only(self):
        t = tree()
        toks = list(range(16))
        assert t.insert(toks, "w0") == 16
        assert t.insert(toks, "w0") == 0
        assert t.insert(toks + list(range(100, 104)), "w0") == 4
        assert t.tenant_token_count["w0"] == 20

    def test_two_tenants_share_nodes(self):
        t = tree()
        toks = list(range(16))
        t.insert(toks, "w0")
        n0 = len(t)
        t.insert(toks, "w1")
        assert len(t) == n0


class TestMatchAndInsert:
    def test_match_resolved_pre_insert(self):
        t = tree()
        toks = list(range(16))
        seen = []
        r, tenant = t.match_and_insert(toks, lambda res: seen.append(res.matched_token_count) or "w0")
        assert seen == [0]  # matched against empty tree
        assert t.match(toks).matched_token_count == 16  # but inserted after

    def test_none_skips_insert(self):
        t = tree()
        t.match_and_insert(list(range(16)), lambda res: None)
        assert len(t) == 0


class TestEviction:
    def test_lru_leaf_eviction(self):
        t = tree()
        t.insert(list(range(16)), "w0")  # older
        t.insert(list(range(100, 116)), "w0")  # newer
        t.evict(max_nodes=4)
        assert len(t) <= 4
        # the newer path survives
        assert t.match(list(range(100, 116))).matched_token_count > 0

    def test_evicted_tenant_blocks_match(self):
        t = tree()
        toks = list(range(8))
        t.insert(toks, "w0")
        t.remove_tenant("w0")
        assert t.match(toks).matched_token_count == 0

    def test_evict_respects_children(self):
        t = tree()
        t.insert(list(range(32)), "w0")  # chain of nodes
        before = len(t)
        t.evict(max_nodes=before)  # no-op
        assert len(t) == before


class TestStringTree:
    def test_byte_prefix(self):
        t = StringTree(page_size=4)
        t.insert_text("hello world, this is a prompt", "w0")
        r = t.match_text("hello world, this is a different")
        assert r.matched_token_count >= 20
        assert r.tenant == "w0"

    def test_unicode_safe(self):
        t = StringTree(page_size=4)
        t.insert_text("héllo wörld ünïcode", "w0")
        r = t.match_text("héllo wörld ünïcode")
        assert r.tenant == "w0"
