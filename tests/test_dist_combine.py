"""Multi-process exchange glue test (CPU, gloo, world_size=2): the exact
allgather-partials + rank-ordered combine that the N>1 bench path performs
over RCCL, validated without a GPU by feeding crafted Jacobian partials."""
import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _worker(rank, world, port, point_hexes, results):
    import sys
    sys.path.insert(0, REPO)
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    import pywrap as oracle
    from spectre_amd import ffi

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        # each rank crafts a distinct partial: its point in window `rank`
        x = bytes([2] + [0] * 31)
        fq_one = oracle.fq_mul(x, oracle.fq_inv(x))
        pt = bytes.fromhex(point_hexes[rank])
        partial = b"".join(
            (pt + fq_one) if w == rank
            else bytes(96) for w in range(ffi.NUM_WINDOWS))
        local = torch.frombuffer(bytearray(partial), dtype=torch.uint8)
        gathered = [torch.zeros_like(local) for _ in range(world)]
        dist.all_gather(gathered, local)
        blob = b"".join(bytes(t.numpy().tobytes()) for t in gathered)
        got = ffi.combine_partials(blob, world)
        results[rank] = got.hex()
    finally:
        dist.destroy_process_group()


def test_allgather_combine_gloo(oracle, golden):
    g1 = golden("g1.json")
    pts = [c["mul"] for c in g1["mul_cases"]
           if bytes.fromhex(c["mul"]) != bytes(64)][:2]
    world = 2
    mgr = mp.Manager()
    results = mgr.dict()
    port = 29571
    mp.spawn(_worker, args=(world, port, pts, results), nprocs=world,
             join=True)
    # expected: P0 * 2^0 + P1 * 2^WINDOW_BITS
    from spectre_amd import ffi
    want = oracle.g1_add(
        bytes.fromhex(pts[0]),
        oracle.g1_mul(bytes.fromhex(pts[1]),
                      (1 << ffi.WINDOW_BITS).to_bytes(32, "little")))
    assert results[0] == results[1] == want.hex()
