"""Multi-GPU sharding logic + the VRF-nonce min-reduce, covered on CPU with
gloo world_size=2 (the N>1 data path of bench.py)."""
import importlib
import os

import pytest

sharding = importlib.import_module("go-spacemesh_amd.sharding")


def test_shard_range_partitions_exactly():
    for total in [1, 7, 64, 2**22, 4 * 2**32]:
        for world in [1, 2, 3, 8]:
            covered = 0
            prev_end = 0
            for r in range(world):
                s, e = sharding.shard_range(total, world, r)
                assert s == prev_end
                prev_end = e
                covered += e - s
            assert prev_end == total
            assert covered == total


def test_merge_nonces_min_label_then_index():
    a = (5, b"\x02" + b"\x00" * 31)
    b = (9, b"\x01" + b"\xff" * 31)
    c = (3, b"\x01" + b"\xff" * 31)  # same label as b, smaller index
    assert sharding.merge_nonces([a, None, b, c]) == c
    assert sharding.merge_nonces([None, None]) is None
    assert sharding.merge_nonces([a]) == a


def _gloo_worker(rank, world, port, nonce_holders):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    # each rank contributes its shard-local candidate; some ranks may have
    # found none (nonce optional per shard — bench.py passes None then)
    if rank in nonce_holders:
        local = (rank * 100 + 1, bytes([rank + 1]) + bytes(31))
    else:
        local = None
    got = sharding.allreduce_nonce(local)
    # the lowest-numbered holder's label is the global minimum
    lo = min(nonce_holders)
    assert got == (lo * 100 + 1, bytes([lo + 1]) + bytes(31)), got
    # the bench's whole-job MAX-over-ranks reduction (bench.py timed region)
    import torch
    t = torch.tensor([float(rank + 1)], dtype=torch.float64)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    assert t.item() == world
    dist.barrier()
    dist.destroy_process_group()


def _run_world(world, port, nonce_holders):
    import torch.multiprocessing as mp
    ctx = mp.get_context("spawn")
    procs = [ctx.Process(target=_gloo_worker,
                         args=(r, world, port, nonce_holders))
             for r in range(world)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(180)
        assert p.exitcode == 0


def test_allreduce_nonce_gloo_world2():
    _run_world(2, 29512, {0, 1})


def test_allreduce_nonce_gloo_world8_sparse():
    """The driver's SCALE shape is 8 ranks; some shards may finish without
    a below-threshold candidate, so the min-reduce must also tolerate
    rank-local None contributions."""
    _run_world(8, 29513, {3, 5})
