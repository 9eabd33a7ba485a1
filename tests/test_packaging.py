"""Packaging: the sdist must carry every subpackage and all native sources
(the reference's CI installs from the sdist to catch exactly this class of
bug — SURVEY.md §4)."""
import subprocess
import sys
import tarfile
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def test_sdist_contains_everything(tmp_path):
    r = subprocess.run(
        [sys.executable, "setup.py", "sdist", "--dist-dir", str(tmp_path)],
        cwd=str(REPO), capture_output=True, text=True, timeout=300,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    (tar_path,) = tmp_path.glob("*.tar.gz")
    names = tarfile.open(tar_path).getnames()
    base = names[0].split("/")[0]

    def has(suffix):
        return any(n.endswith(suffix) for n in names)

    # every subpackage ships
    for sub in ["cli", "config", "data", "models", "ops", "parallel",
                "pipeline", "serve", "train", "utils", "vocab"]:
        assert f"{base}/spacy_ray_amd/{sub}/__init__.py" in names, sub
    # native sources ship; hipify artifacts don't
    assert has("ops/kernels/srx_ext.hip")
    assert has("ops/kernels/srx_common.hip.h")
    assert has("ops/csrc/transitions.cpp")
    assert has("ops/csrc/murmur3.h")
    assert not any(n.endswith("_hip.hip") for n in names)
