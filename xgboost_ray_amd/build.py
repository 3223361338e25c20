"""`python -m xgboost_ray_amd.build` - compile the gfx950 HIP extension."""

import os
import subprocess
import sys

if __name__ == "__main__":
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    sys.exit(
        subprocess.run(
            [sys.executable, "setup.py", "build_ext", "--inplace"],
            cwd=repo,
            env=env,
        ).returncode
    )
