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


def _worker_windows(rank, world, port, point_hexes, results):
    import sys
    sys.path.insert(0, REPO)
    sys.path.insert(0, os.path.join(REPO, "oracle"))
    import pywrap as oracle
    from spectre_amd import ffi

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        # window sharding: rank owns NUM_WINDOWS/world consecutive windows
        # and sends ONLY those (disjoint slices, allgather, no reduction)
        w_cnt = ffi.NUM_WINDOWS // world
        x = bytes([2] + [0] * 31)
        fq_one = oracle.fq_mul(x, oracle.fq_inv(x))
        pt = bytes.fromhex(point_hexes[rank])
        # this rank's slice: its point in the FIRST of its windows
        partial = b"".join(
            (pt + fq_one) if w == 0 else bytes(96) for w in range(w_cnt))
        local = torch.frombuffer(bytearray(partial), dtype=torch.uint8)
        gathered = [torch.zeros_like(local) for _ in range(world)]
        dist.all_gather(gathered, local)
        blob = b"".join(bytes(t.numpy().tobytes()) for t in gathered)
        got = ffi.combine_window_partials(blob, world)
        results[rank] = got.hex()
    finally:
        dist.destroy_process_group()


def test_allgather_combine_windows_gloo(oracle, golden):
    """The window-sharded exchange (disjoint slices, pure allgather) that
    the N>1 bench path performs by default."""
    g1 = golden("g1.json")
    pts = [c["mul"] for c in g1["mul_cases"]
           if bytes.fromhex(c["mul"]) != bytes(64)][:2]
    world = 2
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_windows, args=(world, 29573, pts, results), nprocs=world,
             join=True)
    from spectre_amd import ffi
    # rank 0's point sits in window 0, rank 1's in window NUM_WINDOWS/2
    want = oracle.g1_add(
        bytes.fromhex(pts[0]),
        oracle.g1_mul(bytes.fromhex(pts[1]),
                      (1 << (ffi.WINDOW_BITS * (ffi.NUM_WINDOWS // 2)))
                      .to_bytes(32, "little")))
    assert results[0] == results[1] == want.hex()


def test_allgather_combine_windows_gloo_world4(oracle, golden):
    """Window exchange at world 4 (each rank 4 windows) — the shape the
    round-end 8-GPU scaling run exercises, CPU-validated with gloo."""
    g1 = golden("g1.json")
    pts = [c["mul"] for c in g1["mul_cases"]
           if bytes.fromhex(c["mul"]) != bytes(64)][:4]
    world = 4
    mgr = mp.Manager()
    results = mgr.dict()
    mp.spawn(_worker_windows, args=(world, 29579, pts, results), nprocs=world,
             join=True)
    from spectre_amd import ffi
    want = None
    wb = ffi.WINDOW_BITS
    per = ffi.NUM_WINDOWS // world
    for r in range(world):
        term = oracle.g1_mul(bytes.fromhex(pts[r]),
                             (1 << (wb * per * r)).to_bytes(32, "little"))
        want = term if want is None else oracle.g1_add(want, term)
    for r in range(world):
        assert results[r] == want.hex(), r


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
