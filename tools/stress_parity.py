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
        kind = rng.choice(["msm", "msm_batch", "msm_shard", "msm_windows",
                           "ntt", "gate"])
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
        elif kind == "msm_windows":
            # window-sharded shards (disjoint window ranges, rank order)
            n = rng.randrange(64, 1 << 13)
            nsh = rng.choice([2, 4, 8])
            sc, bs = oracle.gen_msm_inputs(n, rng.randrange(1 << 30), fast=True)
            want = gpu.msm(bs, sc, n)
            d_b = gpu.malloc(64 * n)
            d_s = gpu.malloc(32 * n)
            gpu.upload(d_b, bs)
            gpu.upload(d_s, sc)
            w_cnt = ffi.NUM_WINDOWS // nsh
            blob = b"".join(
                gpu.msm_shard_windows_device(d_b, d_s, n, r * w_cnt, w_cnt)
                for r in range(nsh))
            gpu.free(d_b)
            gpu.free(d_s)
            assert ffi.combine_window_partials(blob, nsh) == want, (kind, n, nsh)
        elif kind == "gate":
            # random well-formed gate program vs bigint evaluation
            log_n = rng.randrange(4, 11)
            n = 1 << log_n
            ncols = rng.randrange(1, 4)
            cols_int = [[rng.randrange(R) for _ in range(n)]
                        for _ in range(ncols)]
            consts = [rng.randrange(R) for _ in range(2)]
            rot_scale = rng.choice([1, 2, 4])
            prog, depth = [], 0
            while len(prog) < rng.randrange(3, 24) or depth != 1:
                if depth >= 2 and (depth >= 7 or rng.random() < 0.5):
                    prog.append((rng.choice([gpu.GATE_ADD, gpu.GATE_SUB,
                                             gpu.GATE_MUL]), 0, 0))
                    depth -= 1
                elif depth >= 1 and rng.random() < 0.1:
                    prog.append((gpu.GATE_NEG, 0, 0))
                elif rng.random() < 0.7:
                    max_rot = max(1, (n - 1) // rot_scale)
                    r_lo = -min(3, max_rot)
                    r_hi = min(3, max_rot)
                    prog.append((gpu.GATE_COL, rng.randrange(ncols),
                                 rng.randrange(r_lo, r_hi + 1)))
                    depth += 1
                else:
                    prog.append((gpu.GATE_CONST, rng.randrange(2), 0))
                    depth += 1
            fr_vec = lambda ints: b"".join(
                oracle.fr_from_canonical(v.to_bytes(32, "little"))
                for v in ints)
            d_cols = []
            for ci in cols_int:
                d = gpu.malloc(32 * n)
                gpu.upload(d, fr_vec(ci))
                d_cols.append(d)
            d_out = gpu.malloc(32 * n)
            gpu.gate_eval(d_cols, fr_vec(consts), prog, n,
                          rot_scale=rot_scale, d_out=d_out)
            raw = bytes(gpu.download(d_out, 32 * n))
            got = [int.from_bytes(
                oracle.fr_to_canonical(raw[32 * i:32 * (i + 1)]), "little")
                for i in range(n)]
            want = []
            for row in range(n):
                st = []
                for op, a_, b_ in prog:
                    if op == gpu.GATE_COL:
                        st.append(cols_int[a_][(row + b_ * rot_scale) % n])
                    elif op == gpu.GATE_CONST:
                        st.append(consts[a_])
                    elif op == gpu.GATE_NEG:
                        st[-1] = (-st[-1]) % R
                    else:
                        rhs = st.pop()
                        lhs = st.pop()
                        st.append((lhs + rhs) % R if op == gpu.GATE_ADD else
                                  (lhs - rhs) % R if op == gpu.GATE_SUB else
                                  lhs * rhs % R)
                want.append(st[0])
            for d in d_cols + [d_out]:
                gpu.free(d)
            assert got == want, (kind, log_n, len(prog))
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
