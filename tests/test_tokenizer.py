import os

from transformer_amd.data import SubwordTokenizer


CORPUS = [
    "he goes to school",
    "the cat sees a dog",
    "one two three four five",
    "the dog is big and the cat is small",
    "she sees the red house",
]


def test_roundtrip():
    tok = SubwordTokenizer.build_from_corpus(CORPUS, target_vocab_size=500)
    for line in CORPUS:
        ids = tok.encode(line)
        assert all(0 < i < tok.vocab_size for i in ids)
        assert tok.decode(ids) == line


def test_oov_bytes_roundtrip():
    tok = SubwordTokenizer.build_from_corpus(CORPUS, target_vocab_size=300)
    s = "zürich blitz 😀"
    assert tok.decode(tok.encode(s)) == s


def test_pad_id_reserved():
    tok = SubwordTokenizer.build_from_corpus(CORPUS, target_vocab_size=300)
    assert 0 not in tok.encode("the cat sees the dog")


def test_save_load(tmp_path):
    tok = SubwordTokenizer.build_from_corpus(CORPUS, target_vocab_size=300)
    prefix = str(tmp_path / "vocab")
    tok.save_to_file(prefix)
    assert os.path.exists(prefix + ".subwords")
    tok2 = SubwordTokenizer.load_from_file(prefix)
    assert tok2.vocab_size == tok.vocab_size
    for line in CORPUS:
        assert tok2.encode(line) == tok.encode(line)


def test_tokenizer_roundtrip_fuzz():
    """Byte fallback must round-trip arbitrary text, including unicode,
    underscores (the end-of-word marker) and OOV symbols."""
    import random

    from transformer_amd.data.tokenizer import SubwordTokenizer

    tok = SubwordTokenizer.build_from_corpus(
        ["the quick brown fox", "pack my box with five dozen jugs"] * 5,
        target_vocab_size=200)
    rng = random.Random(7)
    alphabet = "abc ДЖ中文🙂_\\ xyz0189"
    for _ in range(40):
        s = "".join(rng.choice(alphabet) for _ in range(rng.randrange(1, 40)))
        # round-trip is defined up to whitespace normalization
        want = " ".join(s.split())
        ids = tok.encode(s)
        assert all(0 < i < tok.vocab_size for i in ids)
        assert tok.decode(ids) == want, (s, tok.decode(ids))


def test_tokenizer_roundtrip_hypothesis():
    """Property: decode(encode(text)) recovers text modulo whitespace
    normalization, for arbitrary unicode (byte fallback covers OOV)."""
    from hypothesis import given, settings, strategies as st

    from transformer_amd.data.tokenizer import SubwordTokenizer

    tok = SubwordTokenizer.build_from_corpus(
        ["the quick brown fox", "pack my box with five dozen jugs"], 300)

    @settings(max_examples=200, deadline=None)
    @given(st.text(max_size=40))
    def check(text):
        ids = tok.encode(text)
        assert all(0 < i < tok.vocab_size for i in ids)
        # whitespace runs collapse to single spaces by construction
        expect = " ".join(text.split())
        assert tok.decode(ids) == expect, (text, ids)

    check()
