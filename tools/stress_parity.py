#!/usr/bin/env python3
"""Randomized stress-parity campaign (GPU box): many random MSM/NTT
configurations against the CPU oracle, bit-exact, plus run-to-run and
shard-count determinism. Seeded; prints a summary line. Not part of the
pytest suite (runtime ~minutes) — run ad hoc to deepen parity evidence."""
import os
import random
import sys
import time

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, HERE)
sys.path.insert(0, os.path.join(HERE, "oracle"))
import pywrap as oracle  # noqa: E402
from spectre_amd import SpectreGpu, ffi  # noqa: E402

R = 21888242871839275222246405745257275088548364400416034343698204186575808495617


def main():
    seed = int(sys.argv[1]) if len(sys.argv) > 1 else 20260915
    budget_s = float(sys.argv[2]) if len(sys.argv) > 2 else 300.0
    rng = random.Random(seed)
    gpu = SpectreGpu([0])
    t0 = time.time()
    checks = 0

    g7 = oracle.fr_from_canonical((7).to_bytes(32, "little"))
    w28 = oracle.fr_pow(g7, ((R - 1) >> 28).to_bytes(32, "little"))

    def omega(log_n):
        w = w28
        for _ in range(28 - log_n):
            w = oracle.fr_mul(w, w)
        return w

    while time.time() - t0 < budget_s:
        kind = rng.choice(["msm", "msm_batch", "msm_shard", "ntt"])
        if kind == "msm":
            n = rng.randrange(1, 1 << rng.randrange(4, 17))
            sc, bs = oracle.gen_msm_inputs(n, rng.randrange(1 << 30), fast=True)
            want = oracle.msm(bs, sc, n)
            canon = rng.random() < 0.5
            s = sc if canon else b"".join(
                oracle.fr_from_canonical(sc[32 * i:32 * i + 32])
                for i in range(n))
            got = gpu.msm(bs, s, n, canonical=canon)
            assert got == want, (kind, n, canon)
            assert gpu.msm(bs, s, n, canonical=canon) == want  # run-to-run
        elif kind == "msm_batch":
            n = rng.randrange(16, 6000)
            nb = rng.randrange(1, 9)
            _, bs = oracle.gen_msm_inputs(n, rng.randrange(1 << 30), fast=True)
            scal, singles = b"", []
            for _ in range(nb):
                sc, _ = oracle.gen_msm_inputs(n, rng.randrange(1 << 30),
                                              fast=True)
                scal += sc
                singles.append(oracle.msm(bs, sc, n))
            assert gpu.msm_batch(bs, scal, nb, n) == singles, (kind, n, nb)
        elif kind == "msm_shard":
            n = rng.randrange(64, 1 << 14)
            nsh = rng.randrange(2, 5)
            sc, bs = oracle.gen_msm_inputs(n, rng.randrange(1 << 30), fast=True)
            want = gpu.msm(bs, sc, n)
            parts = b""
            bounds = [n * i // nsh for i in range(nsh + 1)]
            for s0, s1 in zip(bounds, bounds[1:]):
                m = s1 - s0
                d_b = gpu.malloc(max(64 * m, 64))
                d_s = gpu.malloc(max(32 * m, 32))
                gpu.upload(d_b, bs[64 * s0:64 * s1])
                gpu.upload(d_s, sc[32 * s0:32 * s1])
                parts += gpu.msm_shard_device(d_b, d_s, m)
                gpu.free(d_b)
                gpu.free(d_s)
            assert ffi.combine_partials(parts, nsh) == want, (kind, n, nsh)
        else:
            log_n = rng.randrange(1, 19)
            n = 1 << log_n
            a = oracle.gen_fr_vector(n, rng.randrange(1 << 30))
            w = omega(log_n)
            inv = rng.random() < 0.5
            coset = (oracle.fr_from_canonical(
                rng.randrange(2, 1000).to_bytes(32, "little"))
                if rng.random() < 0.5 else None)
            om = oracle.fr_inv(w) if inv else w
            cg = (oracle.fr_inv(coset) if (coset and inv) else coset)
            got = gpu.ntt(a, log_n, om, inverse=inv, coset_gen=cg)
            want = oracle.ntt(a, log_n, om, inverse=inv, coset_gen=cg)
            assert got == want, (kind, log_n, inv, coset is not None)
        checks += 1
    print(f"stress parity OK: {checks} randomized configurations bit-exact "
          f"in {time.time() - t0:.0f}s (seed {seed})")


if __name__ == "__main__":
    main()
