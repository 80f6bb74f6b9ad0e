"""In-tree build of the HIP extension (libsdbv.so) for gfx950.

The .so is built in-tree so it travels with the repo snapshot to GPU boxes
(JIT caches under ~/.cache do not). hipcc cross-compiles on CPU-only hosts.
"""
import os
import subprocess

_DIR = os.path.dirname(os.path.abspath(__file__))
SO_PATH = os.path.join(_DIR, "libsdbv.so")
SRC = os.path.join(_DIR, "csrc", "sdbv.hip")

CMD = [
    "hipcc",
    "--offload-arch=gfx950",
    "-O3",
    "-std=c++17",
    "-fPIC",
    "-shared",
    SRC,
    "-o",
    SO_PATH,
]


def build(force=False):
    if (not force and os.path.exists(SO_PATH)
            and os.path.getmtime(SO_PATH) >= os.path.getmtime(SRC)):
        return SO_PATH
    subprocess.run(CMD, check=True)
    return SO_PATH


if __name__ == "__main__":
    print(build(force=True))
