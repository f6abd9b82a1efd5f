"""Build helpers: compile the product engine (hipcc, gfx950) and the oracle
(gcc). Used by __graft_entry__.build() and tests/conftest.py."""
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _make(path, target=None):
    cmd = ["make", "-C", path, "-j", str(os.cpu_count() or 4)]
    if target:
        cmd.append(target)
    r = subprocess.run(cmd, capture_output=True, text=True)
    if r.returncode != 0:
        sys.stderr.write(r.stdout[-4000:])
        sys.stderr.write(r.stderr[-4000:])
        raise RuntimeError("build failed in %s" % path)


def build_product():
    """libcopr.so: HIP kernels (gfx950) + C++ runner + generator."""
    _make(os.path.join(ROOT, "tikv_amd", "csrc"))


def build_oracle():
    """liboracle.so: CPU restatement (test infrastructure)."""
    _make(os.path.join(ROOT, "oracle"))


def build_all(force=False):
    prod = os.path.join(ROOT, "tikv_amd", "libcopr.so")
    orc = os.path.join(ROOT, "oracle", "liboracle.so")
    if force or not os.path.exists(prod):
        build_product()
    else:
        _make(os.path.join(ROOT, "tikv_amd", "csrc"))  # incremental
    if force or not os.path.exists(orc):
        build_oracle()
    else:
        _make(os.path.join(ROOT, "oracle"))


if __name__ == "__main__":
    build_all(force="--force" in sys.argv)
    print("build ok")
