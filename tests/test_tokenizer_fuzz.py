"""Property-based fuzz of both tokenizers.

- ByteTokenizer (engine/tokenizer.py): greedy word-token encoding must
  round-trip EXACTLY for arbitrary text — word tokens are an engine-side
  compression and may never change the byte stream.
- BpeTokenizer (engine/bpe_tokenizer.py): id-exact parity with the
  `tokenizers` library on arbitrary text for a tokenizer.json trained
  in-session, plus decode round-trips.
"""
from __future__ import annotations

import pytest

hypothesis = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st  # noqa: E402

from runbookai_amd.engine.tokenizer import ACTIVE_VOCAB, WORD_BASE, ByteTokenizer  # noqa: E402

texts = st.text(
    alphabet=st.characters(min_codepoint=0x20, max_codepoint=0x2FA1F,
                           blacklist_categories=("Cs",)),
    min_size=0, max_size=200)
ascii_texts = st.text(
    alphabet=st.characters(min_codepoint=0x20, max_codepoint=0x7E),
    min_size=0, max_size=300)


class TestByteTokenizerFuzz:
    @settings(max_examples=200, deadline=None)
    @given(text=texts)
    def test_roundtrip_exact(self, text):
        tok = ByteTokenizer(128256)
        assert tok.decode(tok.encode(text)) == text

    @settings(max_examples=150, deadline=None)
    @given(text=ascii_texts)
    def test_word_tokens_stay_in_active_vocab(self, text):
        tok = ByteTokenizer(128256)
        ids = tok.encode(text)
        assert all(0 <= i < ACTIVE_VOCAB for i in ids)
        # word tokens only ever substitute for their exact byte expansion
        n_bytes = len(text.encode("utf-8"))
        assert len(ids) <= n_bytes
        if not any(i >= WORD_BASE for i in ids):
            assert len(ids) == n_bytes

    @settings(max_examples=100, deadline=None)
    @given(a=ascii_texts, b=ascii_texts)
    def test_concat_decode(self, a, b):
        """Token streams concatenate the way byte streams do."""
        tok = ByteTokenizer(128256)
        assert tok.decode(tok.encode(a) + tok.encode(b)) == a + b


@pytest.fixture(scope="module")
def trained_bpe(tmp_path_factory):
    tokenizers = pytest.importorskip("tokenizers")
    corpus = [
        "Redis connection pool exhausted on checkout-api",
        "error rate spiked to 40% after deploy 2024-06-01",
        "kubectl get pods -n prod | grep CrashLoopBackOff",
        "The quick brown fox jumps over the lazy dog's tail, twice!",
    ] * 25
    tok = tokenizers.ByteLevelBPETokenizer()
    tok.train_from_iterator(corpus, vocab_size=700, min_frequency=2)
    path = tmp_path_factory.mktemp("bpe_fuzz") / "tokenizer.json"
    tok.save(str(path))
    from runbookai_amd.engine.bpe_tokenizer import BpeTokenizer

    return BpeTokenizer.from_file(str(path)), tokenizers.Tokenizer.from_file(str(path))


class TestBpeTokenizerFuzz:
    @settings(max_examples=200, deadline=None)
    @given(text=texts)
    def test_id_parity_with_tokenizers_lib(self, text, trained_bpe):
        ours, theirs = trained_bpe
        assert ours.encode(text) == theirs.encode(text).ids

    @settings(max_examples=150, deadline=None)
    @given(text=texts)
    def test_decode_roundtrip(self, text, trained_bpe):
        ours, _ = trained_bpe
        assert ours.decode(ours.encode(text)) == text
