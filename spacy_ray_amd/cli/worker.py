"""Worker process entry: ``python -m spacy_ray_amd.cli.worker config.cfg ...``
launched once per rank by the launcher (RANK/WORLD_SIZE etc. in env)."""
from __future__ import annotations

import argparse
import json
import os
from pathlib import Path

from spacy_ray_amd.config.config import Config
from spacy_ray_amd.train.worker import distributed_train


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("config_path", type=Path)
    ap.add_argument("--output", type=Path, default=None)
    ap.add_argument("--code", type=Path, default=None)
    ap.add_argument("--gpu-id", type=int, default=-1)
    ap.add_argument("--resume", action="store_true")
    ap.add_argument("--overrides-json", type=str, default=None)
    args = ap.parse_args()
    overrides = json.loads(args.overrides_json) if args.overrides_json else None
    config = Config.from_disk(args.config_path, overrides=overrides)
    metrics = (args.output / "metrics.jsonl") if (args.output and int(os.environ.get("RANK", "0")) == 0) else None
    if metrics:
        args.output.mkdir(parents=True, exist_ok=True)
    distributed_train(
        config,
        output_path=args.output,
        use_gpu=args.gpu_id,
        code_path=args.code,
        resume=args.resume,
        metrics_path=metrics,
    )


if __name__ == "__main__":
    main()
