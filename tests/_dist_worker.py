"""Subprocess worker for test_dist_cpu (gloo world_size 2)."""
import json
import os
import sys

root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
if root not in sys.path:
    sys.path.insert(0, root)


def main():
    rank = int(sys.argv[1])
    world = int(sys.argv[2])
    port = sys.argv[3]
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = port
    import torch.distributed as dist
    dist.init_process_group("gloo", rank=rank, world_size=world)
    from tikv_amd.dist import (merge_count, merge_checksum, merge_sum_i128,
                               merge_sum_real)

    c = merge_count(100 + rank)
    x = merge_checksum(0xDEAD0000 + rank)
    v = (1 << 70) if rank == 0 else -1
    lo, hi = merge_sum_i128(v & (2**64 - 1), (v >> 64) & (2**64 - 1))
    r = merge_sum_real(1.25 if rank == 0 else -0.5)
    if rank == 0:
        print(json.dumps({"c": c, "x": x, "lo": lo, "hi": hi, "r": r}))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
