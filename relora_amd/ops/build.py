"""In-tree build helper: `python -m relora_amd.ops.build` compiles the HIP
extension for gfx950 (cross-compiles fine on machines without a GPU)."""

import os
import subprocess
import sys


def build(verbose=True):
    repo_root = os.path.abspath(os.path.join(os.path.dirname(__file__), "..", ".."))
    env = dict(os.environ)
    env.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    env.setdefault("MAX_JOBS", str(min(os.cpu_count() or 8, 16)))
    cmd = [sys.executable, "setup.py", "build_ext", "--inplace"]
    res = subprocess.run(cmd, cwd=repo_root, env=env,
                         capture_output=not verbose, text=True)
    if res.returncode != 0:
        if not verbose:
            sys.stderr.write((res.stdout or "") + (res.stderr or ""))
        raise RuntimeError("HIP extension build failed")
    return True


if __name__ == "__main__":
    build()
