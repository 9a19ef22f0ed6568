"""Build the HIP extension in-tree: ``python -m baton_amd.ops.build``.

Cross-compiles for gfx950 (works without a GPU — hipcc only needs the
target arch). The built ``_hip_ops*.so`` lands in ``baton_amd/ops/`` and is
git-ignored but NOT gpurun-ignored, so it travels to GPU boxes with the
repo snapshot.
"""

from __future__ import annotations

import os
import subprocess
import sys


def build(verbose: bool = True) -> str:
    here = os.path.dirname(os.path.abspath(__file__))
    env = dict(os.environ)
    env.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    env.setdefault("MAX_JOBS", str(min(os.cpu_count() or 8, 16)))
    cmd = [sys.executable, "setup.py", "build_ext", "--inplace"]
    proc = subprocess.run(
        cmd, cwd=here, env=env, capture_output=True, text=True
    )
    if proc.returncode != 0:
        sys.stderr.write(proc.stdout[-4000:])
        sys.stderr.write(proc.stderr[-8000:])
        raise RuntimeError("HIP extension build failed")
    if verbose:
        tail = [l for l in proc.stdout.splitlines() if l.strip()][-3:]
        print("\n".join(tail))
    import glob

    sos = glob.glob(os.path.join(here, "_hip_ops*.so"))
    if not sos:
        raise RuntimeError("build reported success but no _hip_ops*.so found")
    return sos[0]


if __name__ == "__main__":
    print("built:", build())
