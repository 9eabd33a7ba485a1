"""Tok2vec pretraining (spaCy's `spacy pretrain` role).

spaCy pretrains tok2vec on raw text with approximate-LM objectives
("characters" / "vectors") and `[initialize] init_tok2vec` then loads the
weights before supervised training.  Offline equivalent here: a masked-
token objective over the murmur-hashed vocabulary — 15% of tokens have
their attr-hash rows replaced by the "[MASK]" word's hashes and a linear
head predicts the ORIGINAL token's NORM-hash bucket from the tok2vec
output (cross-entropy over `n_buckets` classes).  The trained encoder
weights save as `tok2vec.safetensors`, loadable via
``training.init_tok2vec`` (config key or dotted CLI override) or
``spacy-mi ray train --init-tok2vec path``.
"""
from __future__ import annotations

from pathlib import Path
from typing import Iterator

import numpy as np
import torch

from spacy_ray_amd.models.batch import TokenBatch
from spacy_ray_amd.vocab.attrs import extract_attr_hashes


def masked_batches(examples, batch_size: int, mask_rate: float, seed: int):
    """Yield lists of docs of ~batch_size docs (pretraining is raw-text:
    only the words are used)."""
    docs = [eg.reference if hasattr(eg, "reference") else eg
            for eg in examples]
    rng = np.random.RandomState(seed)
    order = rng.permutation(len(docs))
    for i in range(0, len(order), batch_size):
        yield [docs[j] for j in order[i:i + batch_size]]


def pretrain_tok2vec(nlp, corpus, *, steps: int = 1000, batch_docs: int = 64,
                     n_buckets: int = 4096, mask_rate: float = 0.15,
                     learn_rate: float = 1e-3, seed: int = 0,
                     log_every: int = 50, log=print):
    """Train nlp's tok2vec on the corpus with the masked-token objective.
    Returns the per-interval mean losses (callers assert they fall)."""
    t2v_pipe = nlp.tok2vec
    assert t2v_pipe is not None and t2v_pipe.module is not None, \
        "pretrain needs an initialized tok2vec pipe"
    module = t2v_pipe.module
    device = nlp.device
    width = t2v_pipe.width
    head = torch.nn.Linear(width, n_buckets).to(device)
    head = head.to(next(module.parameters()).dtype)
    params = list(module.parameters()) + list(head.parameters())
    opt = torch.optim.Adam(params, lr=learn_rate)
    mask_hashes = torch.from_numpy(
        extract_attr_hashes(["[MASK]"]).view(np.int64)).to(device)  # [1, 4]
    rng = np.random.RandomState(seed)
    examples = list(corpus(nlp))
    losses, interval = [], []
    step = 0
    while step < steps:
        for docs in masked_batches(examples, batch_docs, mask_rate,
                                   seed + step):
            if step >= steps:
                break
            batch = TokenBatch(docs, device)
            total = batch.n_real_tokens
            if total == 0:
                continue
            mask = torch.from_numpy(
                (rng.random_sample(total) < mask_rate)).to(device)
            if not bool(mask.any()):
                continue
            ids = batch.attr_ids.clone()
            # original NORM hash -> target bucket (before masking)
            targets = (ids[:total, 0].view(torch.int64).remainder(n_buckets))
            ids[:total][mask] = mask_hashes[0]
            batch.attr_ids = ids
            t2v = t2v_pipe.forward(batch)
            logits = head(t2v[:total][mask])
            loss = torch.nn.functional.cross_entropy(
                logits.float(), targets[mask])
            opt.zero_grad(set_to_none=True)
            loss.backward()
            opt.step()
            interval.append(float(loss.detach()))
            step += 1
            if step % log_every == 0:
                losses.append(sum(interval) / len(interval))
                log(f"[pretrain] step {step}  loss {losses[-1]:.4f}")
                interval = []
    if interval:
        losses.append(sum(interval) / len(interval))
    return losses


def save_tok2vec(nlp, out_dir) -> Path:
    from safetensors.torch import save_file

    out = Path(out_dir)
    out.mkdir(parents=True, exist_ok=True)
    module = nlp.tok2vec.module
    state = {k: v.detach().cpu().contiguous()
             for k, v in module.state_dict().items()}
    path = out / "tok2vec.safetensors"
    save_file(state, str(path))
    return path


def load_init_tok2vec(nlp, path) -> int:
    """Load pretrained tok2vec weights into the pipeline's tok2vec
    (spaCy's `[initialize] init_tok2vec` contract).  Returns the number of
    tensors loaded; shape mismatches fail loudly."""
    from safetensors.torch import load_file

    p = Path(path)
    if p.is_dir():
        p = p / "tok2vec.safetensors"
    state = load_file(str(p))
    module = nlp.tok2vec.module
    module.load_state_dict(state)
    module.to(nlp.device)
    return len(state)
