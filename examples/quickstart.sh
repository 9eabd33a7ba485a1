#!/usr/bin/env bash
# The full lifecycle on real data (every step is covered by tests/).
# Without `pip install -e .`, substitute `python -m spacy_ray_amd.cli.main`
# for `spacy-mi` (identical surface).
set -euo pipefail

# 0. build the C++ core + gfx950 HIP kernels (in-tree, no network)
python setup.py build_ext --inplace

# 1. generate a config
spacy-mi init config quickstart.cfg --pipeline tagger,parser,ner --arch cnn --gpu

# 2. bring your own data (CoNLL-U or CoNLL-02/03 IOB)
spacy-mi convert train.conllu train.spacy
spacy-mi convert dev.conllu dev.spacy

# 3. pre-flight checks
spacy-mi debug config quickstart.cfg --paths.train train.spacy --paths.dev dev.spacy
spacy-mi debug data quickstart.cfg --paths.train train.spacy --paths.dev dev.spacy

# 4. train on 8 GPUs (one process per GPU, ZeRO-1 over RCCL/xGMI)
spacy-mi ray train quickstart.cfg \
    --n-workers 8 --gpu-id 0 --output ./model \
    --paths.train train.spacy --paths.dev dev.spacy

# 5. evaluate a checkpoint
spacy-mi ray evaluate ./model/model-best --gpu-id 0

# 6. serve it
spacy-mi serve ./model/model-best --gpu-id 0 --port 8000 &
sleep 5
curl -s localhost:8000/annotate -H 'content-type: application/json' \
     -d '{"texts": ["An example sentence to annotate"]}'
