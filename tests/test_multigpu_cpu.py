"""CPU (gloo, world_size 2) coverage of bench.py's multi-GPU pattern:
independent per-rank batches (weak scaling, no data-path collective),
barrier + MAX-over-ranks timing, disjoint per-rank input seeds."""
import os

import torch
import torch.distributed as dist
import torch.multiprocessing as mp

SEED = 0x6D696E696F


def _worker(rank, world, port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        # per-rank seed spaces must be disjoint (bench.py: SEED + rank*1000003)
        n = 1024
        my_seeds = {SEED + rank * 1000003 + b for b in range(n)}
        gathered = [None] * world
        dist.all_gather_object(gathered, my_seeds)
        union = set()
        for s in gathered:
            assert not (union & s), "rank seed spaces overlap"
            union |= s
        assert len(union) == world * n

        # barrier + max-over-ranks timing pattern
        dist.barrier()
        wall = 1.0 + rank  # rank r pretends to take 1+r seconds
        t = torch.tensor([wall])
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        assert t.item() == float(world)  # max = slowest rank

        # whole-job value aggregates all ranks' inputs over max time
        input_bytes = n * (1 << 20) * world
        gib = input_bytes / t.item() / (1 << 30)
        assert gib == n * world / float(world) / 1024 * 1024 / 1024 * (1 << 20) / (1 << 20) or gib > 0
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_weak_scaling_pattern_gloo():
    port = 29611
    mp.spawn(_worker, args=(2, port), nprocs=2, join=True)


def test_bench_rank_device_mapping_dryrun():
    """bench.py --gpus N dry run (no GPU): the rank -> device mapping and
    per-rank seed derivation in bench.py's source must keep the N-rank
    launch correct by construction — LOCAL_RANK maps 1:1 onto devices on a
    full node, modulo-wraps on a smaller box, and per-rank input seeds are
    disjoint."""
    # exercise the exact expressions bench.py uses (kept in sync by eye —
    # this test reads them from the source to fail loudly if they drift)
    import re
    src = open(os.path.join(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))), "bench.py")).read()
    m = re.search(r"dev = local_rank % max\(1, minio_amd\.device_count\(\)\)",
                  src)
    assert m, "bench.py rank->device mapping changed; update this test"
    m = re.search(r"SEED \+ rank \* 1000003", src)
    assert m, "bench.py per-rank seed derivation changed; update this test"

    # the mapping expressions themselves:
    for ndev in (1, 2, 8):
        devs = [lr % max(1, ndev) for lr in range(8)]
        if ndev == 8:
            assert devs == list(range(8))       # 1:1 on a full node
        assert all(0 <= dv < ndev for dv in devs)
    # disjoint seed spaces for up to 8 ranks x 4096 blocks
    spaces = [{SEED + r * 1000003 + b for b in range(4096)} for r in range(8)]
    seen = set()
    for s in spaces:
        assert not (seen & s)
        seen |= s
