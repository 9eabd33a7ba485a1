"""torch.profiler over a few flagship bench steps — host-op diagnosis."""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.abspath(__file__)), ".."))

import torch
from torch.profiler import ProfilerActivity, profile


def main():
    from spacy_ray_amd.config.config import Config, resolve
    from spacy_ray_amd.parallel.comm import LocalComm
    from spacy_ray_amd.parallel.engine import ZeRO1Engine
    from spacy_ray_amd.pipeline.language import init_nlp
    import bench as b

    device = "cuda:0"
    here = os.path.join(os.path.dirname(os.path.abspath(__file__)), "..")
    config = Config.from_disk(os.path.join(here, "examples/configs/en_core_cnn.cfg"))
    nlp = init_nlp(config, device=device, sample_size=64)
    T = resolve(config.interpolate()["training"], validate=False)
    engine = ZeRO1Engine(nlp, T["optimizer"], LocalComm())
    batches = b.build_batches(nlp, batch_words=128000, n_batches=4, seed=1,
                              words_per_doc=20, vocab_size=5000)
    for i in range(3):
        engine.accumulate(batches[i % 4], drop=0.1)
        engine.apply_step()
    torch.cuda.synchronize()
    with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
        for i in range(2):
            engine.accumulate(batches[i % 4], drop=0.1)
            engine.apply_step()
        torch.cuda.synchronize()
    print(prof.key_averages().table(sort_by="self_cpu_time_total", row_limit=25))


if __name__ == "__main__":
    main()
