"""bench.py contract guards that run without a GPU."""
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_gpus_flag_refuses_single_rank_masquerade():
    """--gpus N with WORLD_SIZE unset and fewer visible GPUs must exit
    non-zero (never report n_gpus N for single-rank work)."""
    env = dict(os.environ)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--gpus", "2",
         "--steps", "1", "--warmup", "0", "--rows", "1000"],
        capture_output=True, text=True, env=env, timeout=300)
    assert r.returncode == 2, (r.returncode, r.stderr[-500:])
    assert "refusing" in r.stderr
