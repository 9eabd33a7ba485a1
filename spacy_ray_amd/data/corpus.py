"""Corpus readers: DocBin files + deterministic synthetic corpora.

Registered under ``@readers`` and resolved from ``[corpora.*]`` blocks via
dotted names (`/root/reference/spacy_ray/worker.py:94-95`).  A corpus is a
callable ``corpus(nlp) -> Iterator[Example]``.

The synthetic corpus is the BASELINE data source (no network for datasets —
BASELINE.md): deterministic per seed, Zipf-distributed synthetic vocabulary,
random projective dependency trees, word-correlated tags/deps and BILUO
entity spans, so models have learnable structure for convergence tests.
"""
from __future__ import annotations

import random
from typing import Iterator, List, Optional

import numpy as np

from spacy_ray_amd.config.registry import registry
from spacy_ray_amd.vocab.doc import Doc, Example, Vocab
from .docbin import DocBin


@registry.readers("spacy.Corpus.v1")
def create_docbin_reader(
    path: Optional[str] = None,
    gold_preproc: bool = False,
    max_length: int = 0,
    limit: int = 0,
    augmenter=None,
):
    def corpus(nlp) -> Iterator[Example]:
        if not path:
            return
        docbin = DocBin.from_disk(path, nlp.vocab)
        n = 0
        for doc in docbin.get_docs(nlp.vocab):
            if max_length and len(doc) > max_length:
                continue
            yield Example.from_doc(doc)
            n += 1
            if limit and n >= limit:
                break

    return corpus


def _random_projective_heads(n: int, rng: random.Random) -> List[int]:
    """Random projective tree over [0, n): recursive root splitting."""
    heads = [-1] * n

    def build(lo: int, hi: int, head: int) -> None:
        if lo >= hi:
            return
        r = rng.randrange(lo, hi)
        heads[r] = head
        build(lo, r, r)
        build(r + 1, hi, r)

    build(0, n, -1)
    return heads


def make_synthetic_docs(
    vocab: Vocab,
    *,
    n_docs: int,
    words_per_doc: int = 20,
    vocab_size: int = 5000,
    n_tags: int = 50,
    n_deps: int = 40,
    n_ent_types: int = 4,
    seed: int = 0,
    world_seed: int = 0,
    tree_style: str = "random",
) -> List[Doc]:
    """`world_seed` fixes the synthetic language itself (word->tag/dep/ent
    mappings) so train/dev corpora with different `seed`s sample different
    docs from the SAME learnable world.  tree_style: "random" projective
    trees, or "chain" (head = previous token; fully learnable — used by
    convergence tests)."""
    rng = random.Random(seed)
    lexicon = [f"w{i}" for i in range(vocab_size)]
    # per-word attr hashes computed ONCE over the lexicon; docs index into
    # this matrix (the per-token string path was the dominant corpus-build
    # cost at bench scales)
    from spacy_ray_amd.vocab.attrs import extract_attr_hashes

    lex_attr = extract_attr_hashes(lexicon)
    # Zipf sampling over the lexicon
    ranks = np.arange(1, vocab_size + 1, dtype=np.float64)
    probs = 1.0 / ranks
    probs /= probs.sum()
    world_rng = np.random.RandomState(world_seed)
    tag_of_word = world_rng.randint(0, n_tags, size=vocab_size)
    dep_of_word = world_rng.randint(0, n_deps, size=vocab_size)
    ent_of_word = world_rng.randint(0, n_ent_types, size=vocab_size)
    np_rng = np.random.RandomState(seed)

    docs: List[Doc] = []
    for _ in range(n_docs):
        n = max(2, int(np_rng.poisson(words_per_doc)))
        word_ids = np_rng.choice(vocab_size, size=n, p=probs)
        words = [lexicon[i] for i in word_ids]
        # tags: word-correlated with 10% noise
        tags = []
        for wid in word_ids:
            t = tag_of_word[wid] if rng.random() > 0.1 else rng.randrange(n_tags)
            tags.append(f"TAG{t}")
        if tree_style == "chain":
            heads = [-1] + list(range(n - 1))
        else:
            heads = _random_projective_heads(n, rng)
        deps = [f"dep{dep_of_word[wid]}" for wid in word_ids]
        for i, h in enumerate(heads):
            if h == -1:
                deps[i] = "ROOT"
        # BILUO entity spans: word-DETERMINISTIC (span starts iff the word id
        # is 0 mod 8; length and type derive from the id) so the pattern is
        # learnable — convergence tests rely on this; ~1 entity per 8 tokens
        ents = ["O"] * n
        i = 0
        while i < n:
            wid = int(word_ids[i])
            if wid % 8 == 0:
                length = min(1 + (wid // 8) % 3, n - i)
                etype = f"ENT{ent_of_word[wid]}"
                if length == 1:
                    ents[i] = f"U-{etype}"
                else:
                    ents[i] = f"B-{etype}"
                    for j in range(i + 1, i + length - 1):
                        ents[j] = f"I-{etype}"
                    ents[i + length - 1] = f"L-{etype}"
                i += length
            else:
                i += 1
        # doc-level category: majority vote of per-word classes (learnable
        # from the pooled representation) + sentence starts every ~7 tokens
        # (word-deterministic: wid % 7 == 0) for senter/textcat pipes
        cat_votes = np.bincount(ent_of_word[word_ids], minlength=n_ent_types)
        cats = {f"CAT{j}": 0.0 for j in range(n_ent_types)}
        cats[f"CAT{int(cat_votes.argmax())}"] = 1.0
        sent_starts = (word_ids % 7 == 0).astype(np.int32)
        sent_starts[0] = 1
        docs.append(Doc(vocab, words, tags=tags, heads=heads, deps=deps,
                        ents=ents, cats=cats, sent_starts=sent_starts,
                        attr_hashes=lex_attr[word_ids]))
    return docs


@registry.readers("spacy-mi.MixedTreebankCorpus.v1")
def create_mixed_treebank_corpus(
    n_treebanks: int = 8,
    docs_per_treebank: int = 500,
    words_per_doc: int = 18,
    vocab_size_per_treebank: int = 4000,
    n_tags: int = 17,
    n_deps: int = 37,
    seed: int = 0,
    shuffle: bool = True,
):
    """Multilingual UD-style stress corpus (BASELINE config #5): N disjoint
    synthetic lexicons (one per 'treebank'/language) sharing one UPOS/dep
    space, mixed into common batches — Zipf-hot rows from several lexicons
    hammer the HashEmbed tables at once."""
    cache: dict = {}

    def corpus(nlp) -> Iterator[Example]:
        if "docs" not in cache:
            docs: List[Doc] = []
            for tb in range(n_treebanks):
                tb_docs = make_synthetic_docs(
                    nlp.vocab,
                    n_docs=docs_per_treebank,
                    words_per_doc=words_per_doc,
                    vocab_size=vocab_size_per_treebank,
                    n_tags=n_tags,
                    n_deps=n_deps,
                    n_ent_types=1,
                    seed=seed * 1000 + tb,
                    world_seed=tb,  # each treebank is its own learnable world
                )
                # language-prefix the word forms so lexicons are disjoint
                for d in tb_docs:
                    docs.append(
                        Doc(nlp.vocab, [f"l{tb}:{w}" for w in d.words],
                            tags=d.tags, heads=d.heads, deps=d.deps, ents=d.ents)
                    )
            cache["docs"] = docs
            cache["epoch"] = 0
        docs = list(cache["docs"])
        if shuffle:
            random.Random(seed + cache["epoch"]).shuffle(docs)
        cache["epoch"] += 1
        for doc in docs:
            yield Example.from_doc(doc)

    return corpus


@registry.readers("spacy-mi.SyntheticCorpus.v1")
def create_synthetic_corpus(
    n_docs: int = 1000,
    words_per_doc: int = 20,
    vocab_size: int = 5000,
    n_tags: int = 50,
    n_deps: int = 40,
    n_ent_types: int = 4,
    seed: int = 0,
    world_seed: int = 0,
    shuffle: bool = True,
):
    cache: dict = {}

    def corpus(nlp) -> Iterator[Example]:
        if "docs" not in cache:
            cache["docs"] = make_synthetic_docs(
                nlp.vocab,
                n_docs=n_docs,
                words_per_doc=words_per_doc,
                vocab_size=vocab_size,
                n_tags=n_tags,
                n_deps=n_deps,
                n_ent_types=n_ent_types,
                seed=seed,
                world_seed=world_seed,
            )
            cache["epoch"] = 0
        docs = list(cache["docs"])
        if shuffle:
            random.Random(seed + cache["epoch"]).shuffle(docs)
        cache["epoch"] += 1
        for doc in docs:
            yield Example.from_doc(doc)

    return corpus
