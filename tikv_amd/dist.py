"""Inter-rank merge steps for Region-sharded execution.

The path shards by Region with no data-path collective; these are the only
exchange steps: the final partial-aggregate merge and the checksum XOR fold
(RCCL over xGMI on GPU nodes — backend "nccl" IS RCCL on ROCm; gloo in CPU
tests). Payloads are KB-scale, latency-bound."""
import torch
import torch.distributed as dist


def _dev():
    return "cuda" if dist.get_backend() == "nccl" else "cpu"


def merge_count(count: int) -> int:
    """sum of per-rank counts (count(*) / count(col) final merge)."""
    t = torch.tensor([count], dtype=torch.long, device=_dev())
    dist.all_reduce(t, op=dist.ReduceOp.SUM)
    return int(t.item())


def merge_checksum(xor_val: int) -> int:
    """CRC64 running XOR is order-independent (checksum.rs:78-87): RCCL has
    no XOR reduce op, so all_gather world-size u64s and fold on host."""
    t = torch.tensor([xor_val], dtype=torch.long, device=_dev())
    outs = [torch.zeros_like(t) for _ in range(dist.get_world_size())]
    dist.all_gather(outs, t)
    acc = 0
    for o in outs:
        acc ^= int(o.item()) & (2**64 - 1)
    return acc


def merge_sum_real(x):
    """all-reduce an f64 partial sum (real aggregates across Region
    shards); parallel order keeps it in the 1-ULP class like the kernels."""
    t = torch.tensor([x], dtype=torch.float64, device=_dev())
    dist.all_reduce(t)
    return float(t.item())


def merge_sum_i128(lo: int, hi: int):
    """elementwise merge of a two's-complement i128 partial sum
    (Decimal partial aggregates travel as scaled i128 limbs)."""
    def signed(x):
        return x - (1 << 64) if x >= (1 << 63) else x
    t = torch.tensor([signed(lo), signed(hi)], dtype=torch.long, device=_dev())
    # sum limbs in int64 with manual carry: gather then fold exactly on host
    outs = [torch.zeros_like(t) for _ in range(dist.get_world_size())]
    dist.all_gather(outs, t)
    total = 0
    for o in outs:
        lo_i = int(o[0].item()) & (2**64 - 1)
        hi_i = int(o[1].item())
        total += (hi_i << 64) + lo_i
    return total & (2**64 - 1), (total >> 64) & (2**64 - 1)
