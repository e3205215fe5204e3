"""L1 prefix-cache correctness: cached-prefix encodes must be token-identical
to full encodes (whitespace-boundary composition; reference cache/l1.rs)."""
import pytest

tokenizers = pytest.importorskip("tokenizers")

from smg_amd.tokenizer.l1_cache import L1CachedTokenizer, L1PrefixCache


@pytest.fixture(scope="module")
def hf_tok(tmp_path_factory):
    from tokenizers import Tokenizer
    from tokenizers.models import BPE
    from tokenizers.pre_tokenizers import ByteLevel
    from tokenizers.decoders import ByteLevel as DecBL
    from tokenizers.trainers import BpeTrainer

    tk = Tokenizer(BPE())
    tk.pre_tokenizer = ByteLevel(add_prefix_space=False)
    tk.decoder = DecBL()
    corpus = ["the quick brown fox jumps over the lazy dog " * 20] * 100
    tk.train_from_iterator(corpus, BpeTrainer(vocab_size=600, initial_alphabet=ByteLevel.alphabet()))

    class W:
        name = "hf"
        vocab_size = tk.get_vocab_size()

        def encode(self, t):
            return tk.encode(t, add_special_tokens=False).ids

        def decode(self, ids):
            return tk.decode(ids)

    return W()


def test_composition_exact(hf_tok):
    l1 = L1CachedTokenizer(hf_tok, store_threshold=100)
    prefix = "the quick brown fox jumps over the lazy dog " * 12  # > 256 chars
    a = prefix + "first unique suffix here"
    b = prefix + "second, totally different ending!"
    full_a = hf_tok.encode(a)
    full_b = hf_tok.encode(b)
    assert l1.encode(a) == full_a  # miss path + store
    assert l1.l1.hits == 0
    assert l1.encode(b) == full_b  # hit path must compose identically
    assert l1.l1.hits == 1


def test_short_text_not_cached(hf_tok):
    l1 = L1CachedTokenizer(hf_tok, store_threshold=100)
    assert l1.encode("short") == hf_tok.encode("short")
    assert l1.l1._entries == {}


def test_memory_eviction(hf_tok):
    l1 = L1CachedTokenizer(hf_tok, max_memory=2000, store_threshold=100)
    for i in range(10):
        l1.encode(f"prompt number {i} " * 40)
    assert l1.l1._memory <= 2000


def test_safe_cut_alignment():
    c = L1PrefixCache()
    text = "x" * 300 + " " + "y" * 100
    cut = c._safe_cut(text, 256)
    assert cut is None or text[cut].isspace()
