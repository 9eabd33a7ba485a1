"""Rule-based tokenizer: spaCy's prefix/suffix/infix algorithm.

Behavioral contract of spaCy's Cython tokenizer (SURVEY.md §2.2 N5,
upstream spacy/tokenizer.pyx — re-implemented from the documented
algorithm, not translated): text splits on whitespace; each chunk is
processed by (1) special-case lookup, (2) token_match (URL-like wholes),
(3) iterative prefix stripping, (4) iterative suffix stripping — both
re-checking specials/token_match after each strip — then (5) infix
splitting of the remainder.  Trailing whitespace attaches to the previous
token's `space` flag (doc.text round-trips).

Training from pre-annotated DocBin corpora never tokenizes (tokenization
is off the hot path); this binds the `spacy-mi serve` / nlp(text)
surface.  Rules serialize to the spaCy-layout `tokenizer` file (msgpack of
the pattern strings + specials, like spaCy's tokenizer.to_bytes).
"""
from __future__ import annotations

import re
from typing import Dict, List, Optional, Sequence, Tuple

import msgpack

# English-flavoured defaults (subset of spaCy's punctuation.py rules,
# rebuilt: quotes/brackets/currency prefixes, punctuation/quote/bracket
# suffixes incl. multi-char ellipses, hyphen/slash/symbol infixes).
DEFAULT_PREFIXES = [
    r"\(", r"\)", r"\[", r"\]", r"\{", r"\}", r"<", r">",
    r"\$", r"£", r"€", r"¥", r"#", r"§", r"%",
    r'"', r"'", r"''", r"``", r"`", r"«", r"»", r"“", r"”", r"‘", r"’",
    r"\.\.\.", r"…", r"--", r"—", r"–", r",", r";", r":", r"!", r"\?",
]
DEFAULT_SUFFIXES = [
    r"\.\.\.", r"…", r"'s", r"'S", r"’s", r"’S", r"''", r"``",
    r"\(", r"\)", r"\[", r"\]", r"\{", r"\}", r"<", r">",
    r'"', r"'", r"`", r"«", r"»", r"“", r"”", r"‘", r"’",
    r",", r";", r":", r"!", r"\?", r"%", r"\$", r"£", r"€", r"¥",
    r"--", r"—", r"–",
    r"(?<=[0-9])\+", r"(?<=[^0-9])\.",
]
DEFAULT_INFIXES = [
    r"\.\.\.", r"…", r"--", r"—", r"–",
    r"(?<=[a-zA-Z])[,!?](?=[a-zA-Z])",
    r"(?<=[a-zA-Z0-9])[:<>=/](?=[a-zA-Z])",
    r"(?<=[a-zA-Z])(?:--|[-—–~])(?=[a-zA-Z])",
]
# spaCy-style English special cases (tokenizer_exceptions subset)
DEFAULT_SPECIALS: Dict[str, List[str]] = {
    "don't": ["do", "n't"], "Don't": ["Do", "n't"],
    "doesn't": ["does", "n't"], "Doesn't": ["Does", "n't"],
    "didn't": ["did", "n't"], "can't": ["ca", "n't"], "Can't": ["Ca", "n't"],
    "cannot": ["can", "not"], "won't": ["wo", "n't"], "Won't": ["Wo", "n't"],
    "isn't": ["is", "n't"], "aren't": ["are", "n't"], "wasn't": ["was", "n't"],
    "weren't": ["were", "n't"], "haven't": ["have", "n't"],
    "hasn't": ["has", "n't"], "hadn't": ["had", "n't"],
    "wouldn't": ["would", "n't"], "couldn't": ["could", "n't"],
    "shouldn't": ["should", "n't"], "mustn't": ["must", "n't"],
    "I'm": ["I", "'m"], "I've": ["I", "'ve"], "I'll": ["I", "'ll"],
    "I'd": ["I", "'d"], "you're": ["you", "'re"], "You're": ["You", "'re"],
    "you've": ["you", "'ve"], "you'll": ["you", "'ll"], "you'd": ["you", "'d"],
    "he's": ["he", "'s"], "He's": ["He", "'s"], "she's": ["she", "'s"],
    "She's": ["She", "'s"], "it's": ["it", "'s"], "It's": ["It", "'s"],
    "we're": ["we", "'re"], "We're": ["We", "'re"], "we've": ["we", "'ve"],
    "we'll": ["we", "'ll"], "they're": ["they", "'re"],
    "they've": ["they", "'ve"], "they'll": ["they", "'ll"],
    "that's": ["that", "'s"], "That's": ["That", "'s"],
    "what's": ["what", "'s"], "What's": ["What", "'s"],
    "who's": ["who", "'s"], "let's": ["let", "'s"], "Let's": ["Let", "'s"],
    "Mr.": ["Mr."], "Mrs.": ["Mrs."], "Ms.": ["Ms."], "Dr.": ["Dr."],
    "St.": ["St."], "vs.": ["vs."], "etc.": ["etc."], "e.g.": ["e.g."],
    "i.e.": ["i.e."], "U.S.": ["U.S."], "U.K.": ["U.K."],
}
DEFAULT_TOKEN_MATCH = r"""(?:https?://|www\.)\S+|[\w.+-]+@[\w-]+\.[\w.-]+"""


class Tokenizer:
    def __init__(
        self,
        prefixes: Optional[Sequence[str]] = None,
        suffixes: Optional[Sequence[str]] = None,
        infixes: Optional[Sequence[str]] = None,
        specials: Optional[Dict[str, List[str]]] = None,
        token_match: Optional[str] = None,
    ) -> None:
        self.prefix_patterns = list(prefixes if prefixes is not None else DEFAULT_PREFIXES)
        self.suffix_patterns = list(suffixes if suffixes is not None else DEFAULT_SUFFIXES)
        self.infix_patterns = list(infixes if infixes is not None else DEFAULT_INFIXES)
        self.specials = dict(specials if specials is not None else DEFAULT_SPECIALS)
        self.token_match_pattern = (token_match if token_match is not None
                                    else DEFAULT_TOKEN_MATCH)
        self._compile()

    def _compile(self) -> None:
        self._prefix_re = re.compile("|".join(f"(?:{p})" for p in self.prefix_patterns)) \
            if self.prefix_patterns else None
        self._suffix_re = re.compile("(?:" + "|".join(f"(?:{p})" for p in self.suffix_patterns) + ")$") \
            if self.suffix_patterns else None
        self._infix_re = re.compile("|".join(f"(?:{p})" for p in self.infix_patterns)) \
            if self.infix_patterns else None
        self._token_match_re = (re.compile(self.token_match_pattern)
                                if self.token_match_pattern else None)

    def add_special_case(self, string: str, tokens: List[str]) -> None:
        assert "".join(tokens) == string.replace(" ", ""), (string, tokens)
        self.specials[string] = list(tokens)

    # ------------------------------------------------------------ algorithm
    def _split_chunk(self, chunk: str) -> List[str]:
        if chunk in self.specials:
            return list(self.specials[chunk])
        if self._token_match_re is not None and self._token_match_re.fullmatch(chunk):
            return [chunk]
        prefixes: List[str] = []
        suffixes: List[str] = []
        word = chunk
        while word:
            if word in self.specials or (
                self._token_match_re is not None
                and self._token_match_re.fullmatch(word)
            ):
                break
            m = self._prefix_re.match(word) if self._prefix_re else None
            if m and m.end() > 0 and m.end() < len(word):
                prefixes.append(word[: m.end()])
                word = word[m.end():]
                continue
            m = self._suffix_re.search(word) if self._suffix_re else None
            if m and m.start() > 0:
                suffixes.append(word[m.start():])
                word = word[: m.start()]
                continue
            break
        middle: List[str] = []
        if word in self.specials:
            middle = list(self.specials[word])
        elif word:
            if self._infix_re is not None:
                last = 0
                for m in self._infix_re.finditer(word):
                    if m.start() == 0 or m.end() == len(word) or m.start() == m.end():
                        continue  # edge infixes belong to prefix/suffix rules
                    middle.append(word[last : m.start()])
                    middle.append(word[m.start() : m.end()])
                    last = m.end()
                middle.append(word[last:])
                middle = [t for t in middle if t]
            else:
                middle = [word]
        return prefixes + middle + list(reversed(suffixes))

    def tokenize(self, text: str) -> Tuple[List[str], List[bool]]:
        """-> (words, trailing-space flags); ''.join interleaved == text
        modulo runs of whitespace collapsing to single spaces."""
        words: List[str] = []
        spaces: List[bool] = []
        for m in re.finditer(r"\S+", text):
            toks = self._split_chunk(m.group())
            has_trailing = m.end() < len(text)
            for i, t in enumerate(toks):
                words.append(t)
                spaces.append(has_trailing if i == len(toks) - 1 else False)
        return words, spaces

    def __call__(self, vocab, text: str):
        from .doc import Doc

        words, spaces = self.tokenize(text)
        return Doc(vocab, words, spaces=spaces)

    # ---------------------------------------------------------- serialization
    def to_bytes(self) -> bytes:
        return msgpack.packb({
            "prefixes": self.prefix_patterns,
            "suffixes": self.suffix_patterns,
            "infixes": self.infix_patterns,
            "specials": self.specials,
            "token_match": self.token_match_pattern,
        }, use_bin_type=True)

    @classmethod
    def from_bytes(cls, data: bytes) -> "Tokenizer":
        msg = msgpack.unpackb(data, raw=False)
        return cls(
            prefixes=msg.get("prefixes"),
            suffixes=msg.get("suffixes"),
            infixes=msg.get("infixes"),
            specials=msg.get("specials"),
            token_match=msg.get("token_match"),
        )


def tokenizer_to_bytes(tok: Optional[Tokenizer]) -> bytes:
    return (tok or Tokenizer()).to_bytes()


def tokenizer_from_bytes(data: bytes) -> Tokenizer:
    return Tokenizer.from_bytes(data)
