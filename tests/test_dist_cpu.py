"""Multi-process merge-step coverage on CPU (gloo, world_size 2)."""
import json
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.abspath(__file__))


def test_merge_steps_gloo():
    worker = os.path.join(HERE, "_dist_worker.py")
    port = "29531"
    procs = [subprocess.Popen([sys.executable, worker, str(r), "2", port],
                              stdout=subprocess.PIPE, text=True)
             for r in range(2)]
    outs = [p.communicate(timeout=180) for p in procs]
    for p in procs:
        assert p.returncode == 0
    res = json.loads(outs[0][0].strip().splitlines()[-1])
    assert res["c"] == 201
    assert res["x"] == (0xDEAD0000 ^ 0xDEAD0001)
    total = (1 << 70) - 1
    assert res["lo"] == total & (2**64 - 1)
    assert res["hi"] == (total >> 64) & (2**64 - 1)
    assert res["r"] == 0.75
