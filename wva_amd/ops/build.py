"""Build the HIP extension in-tree: `python -m wva_amd.ops.build`."""
from __future__ import annotations

import os
import subprocess
import sys


def build() -> None:
    here = os.path.dirname(os.path.abspath(__file__))
    env = dict(os.environ)
    env.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    env.setdefault("MAX_JOBS", "8")
    subprocess.check_call(
        [sys.executable, "setup.py", "build_ext", "--inplace"],
        cwd=here,
        env=env,
    )


if __name__ == "__main__":
    build()
