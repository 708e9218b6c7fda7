#!/usr/bin/env python3
"""Golden-vector generator for the BN254 MSM/NTT hot path.

Run from the repo root:  python3 tests/golden/generate.py
Writes JSON fixtures next to itself. Deterministic (fixed seeds, SplitMix64).

The fixtures pin the C oracle (oracle/) — and transitively the HIP kernels,
which are parity-tested against the oracle — to an independent Python-bigint
restatement of the reference algorithms (see bn254_ref.py header for the
reference citations). Inputs for large cases are seed-derived (SplitMix64,
spec below) so fixtures stay small; expected outputs are stored in full for
point-valued results and as SHA-256 digests for large vectors.

SplitMix64 (must match oracle/bn254.c:splitmix64 exactly):
    z += 0x9E3779B97F4A7C15; x = z
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9   (mod 2^64)
    x = (x ^ (x >> 27)) * 0x94D049BB133111EB   (mod 2^64)
    return x ^ (x >> 31)
An Fr draw takes 4 consecutive u64s, assembles a 256-bit LE integer, and
reduces mod r. An Fq draw is the same mod p.
"""
import hashlib
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from bn254_ref import (  # noqa: E402
    P, R, G1_GEN, G2_GEN, fr_root_of_unity, to_mont_bytes, to_canon_bytes,
    g1_to_bytes, g1_add, g1_mul, g1_neg, g1_is_on_curve, msm, ntt, intt,
    distribute_powers, g2_to_bytes, g2_add, g2_mul, g2_neg, g2_msm,
    g2_is_on_curve,
)

HERE = os.path.dirname(os.path.abspath(__file__))
M64 = (1 << 64) - 1


class SplitMix64:
    def __init__(self, seed: int):
        self.z = seed & M64

    def next_u64(self) -> int:
        self.z = (self.z + 0x9E3779B97F4A7C15) & M64
        x = self.z
        x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & M64
        x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & M64
        return x ^ (x >> 31)

    def next_int256(self) -> int:
        v = 0
        for i in range(4):
            v |= self.next_u64() << (64 * i)
        return v

    def next_fr(self) -> int:
        return self.next_int256() % R

    def next_fq(self) -> int:
        return self.next_int256() % P


def hx(b: bytes) -> str:
    return b.hex()


def sha(b: bytes) -> str:
    return hashlib.sha256(b).hexdigest()


# ------------------------------------------------------------ field vectors
def gen_field(modulus, name):
    rng = SplitMix64(0xF1E1D + (0 if name == "fr" else 1))
    cases = []
    specials = [0, 1, 2, modulus - 1, modulus - 2, (1 << 253) % modulus]
    vals = specials + [rng.next_int256() % modulus for _ in range(10)]
    for i in range(0, len(vals) - 1):
        a, b = vals[i], vals[i + 1]
        cases.append({
            "a_mont": hx(to_mont_bytes(a, modulus)),
            "b_mont": hx(to_mont_bytes(b, modulus)),
            "a_canon": hx(to_canon_bytes(a)),
            "add": hx(to_mont_bytes((a + b) % modulus, modulus)),
            "sub": hx(to_mont_bytes((a - b) % modulus, modulus)),
            "mul": hx(to_mont_bytes(a * b % modulus, modulus)),
            "sqr": hx(to_mont_bytes(a * a % modulus, modulus)),
            "inv": hx(to_mont_bytes(pow(a, -1, modulus) if a else 0, modulus)),
        })
    return {"modulus": hex(modulus), "cases": cases}


# ------------------------------------------------------------ G1 vectors
def gen_g1():
    rng = SplitMix64(0x61AF)
    G = G1_GEN
    pts = [None, G, g1_add(G, G), g1_mul(G, rng.next_fr()), g1_mul(G, rng.next_fr())]
    cases = []
    for a in pts:
        for b in pts:
            cases.append({
                "a": hx(g1_to_bytes(a)), "b": hx(g1_to_bytes(b)),
                "add": hx(g1_to_bytes(g1_add(a, b))),
            })
        cases.append({"a": hx(g1_to_bytes(a)), "b": hx(g1_to_bytes(g1_neg(a))),
                      "add": hx(g1_to_bytes(None))})
    muls = []
    for k in [0, 1, 2, 3, R - 1, R - 2, (1 << 253) - 1, rng.next_fr()]:
        muls.append({"p": hx(g1_to_bytes(G)), "k": hx(to_canon_bytes(k % R)),
                     "mul": hx(g1_to_bytes(g1_mul(G, k)))})
        Q = g1_mul(G, 0xDEADBEEF)
        muls.append({"p": hx(g1_to_bytes(Q)), "k": hx(to_canon_bytes(k % R)),
                     "mul": hx(g1_to_bytes(g1_mul(Q, k)))})
    return {"add_cases": cases, "mul_cases": muls}


# ------------------------------------------------------------ MSM vectors
def seeded_msm_inputs(n, seed):
    """The derivation contract shared with oracle tests: interleaved draws —
    for i in 0..n: scalar_i = next_fr(); base_i = next_fr()*G."""
    rng = SplitMix64(seed)
    scalars, points = [], []
    for _ in range(n):
        scalars.append(rng.next_fr())
        points.append(g1_mul(G1_GEN, rng.next_fr()))
    return scalars, points


def gen_msm():
    out = {"inline_cases": [], "seeded_cases": []}
    G = G1_GEN
    P2 = g1_mul(G, 7)
    P3 = g1_mul(G, 0x123456789ABCDEF)
    inline = [
        ([], []),
        ([0], [G]),
        ([1], [G]),
        ([R - 1], [G]),
        ([5], [None]),                       # identity base
        ([0, 0, 0], [G, P2, P3]),            # all-zero scalars
        ([R - 1, R - 1], [G, g1_neg(G)]),    # cancellation -> identity
        ([3, 3, 3], [G, G, G]),              # duplicate points
        ([1 << 253, (1 << 253) - 1, R - 1, 1, 0], [G, P2, P3, g1_neg(P2), G]),
        ([0xFFFF * sum(1 << (16 * i) for i in range(15))] , [G]),  # max window digits
    ]
    rng = SplitMix64(0x5EED)
    for n in [2, 3, 8, 17]:
        scalars = [rng.next_fr() for _ in range(n)]
        points = [g1_mul(G, rng.next_fr()) for _ in range(n)]
        inline.append((scalars, points))
    for scalars, points in inline:
        out["inline_cases"].append({
            "n": len(scalars),
            "scalars_canon": [hx(to_canon_bytes(s)) for s in scalars],
            "scalars_mont": [hx(to_mont_bytes(s, R)) for s in scalars],
            "bases": [hx(g1_to_bytes(p)) for p in points],
            "result": hx(g1_to_bytes(msm(scalars, points))),
        })
    for n, seed in [(64, 11), (257, 12), (1000, 13)]:
        scalars, points = seeded_msm_inputs(n, seed)
        out["seeded_cases"].append({
            "n": n, "seed": seed,
            "result": hx(g1_to_bytes(msm(scalars, points))),
        })
    return out


# ------------------------------------------------------------ NTT vectors
def seeded_fr_vector(n, seed):
    rng = SplitMix64(seed)
    return [rng.next_fr() for _ in range(n)]


def gen_ntt():
    out = {"inline_cases": [], "seeded_cases": []}
    for log_n in [0, 1, 2, 3, 6, 8]:
        n = 1 << log_n
        omega = fr_root_of_unity(log_n)
        a = seeded_fr_vector(n, 0x77 + log_n)
        fwd = ntt(a, omega, log_n)
        inv = intt(a, omega, log_n)
        g = 5  # arbitrary coset generator for the fixture
        coset_fwd = ntt(distribute_powers(a, g), omega, log_n)
        out["inline_cases"].append({
            "log_n": log_n,
            "omega_mont": hx(to_mont_bytes(omega, R)),
            "coset_g_mont": hx(to_mont_bytes(g, R)),
            "input_mont": [hx(to_mont_bytes(x, R)) for x in a],
            "ntt": [hx(to_mont_bytes(x, R)) for x in fwd],
            "intt": [hx(to_mont_bytes(x, R)) for x in inv],
            "coset_ntt": [hx(to_mont_bytes(x, R)) for x in coset_fwd],
        })
    for log_n, seed in [(10, 0x100), (12, 0x120)]:
        n = 1 << log_n
        omega = fr_root_of_unity(log_n)
        a = seeded_fr_vector(n, seed)
        fwd = ntt(a, omega, log_n)
        inv = intt(a, omega, log_n)
        out["seeded_cases"].append({
            "log_n": log_n, "seed": seed,
            "omega_mont": hx(to_mont_bytes(omega, R)),
            "ntt_sha256": sha(b"".join(to_mont_bytes(x, R) for x in fwd)),
            "intt_sha256": sha(b"".join(to_mont_bytes(x, R) for x in inv)),
        })
    return out


# ------------------------------------------------------------ G2 vectors
def gen_g2():
    """G2 (Fq2 twist) — SURVEY §8a minor row: the reference's G2 MSMs are
    the size <= 2 SRS/verifier ones; the oracle covers the algebra and these
    fixtures pin it to this independent bigint restatement."""
    rng = SplitMix64(0x62AF)
    G = G2_GEN
    assert g2_is_on_curve(G)
    pts = [None, G, g2_add(G, G), g2_mul(G, rng.next_fr()), g2_mul(G, rng.next_fr())]
    cases = []
    for a in pts:
        for b in pts:
            cases.append({"a": hx(g2_to_bytes(a)), "b": hx(g2_to_bytes(b)),
                          "add": hx(g2_to_bytes(g2_add(a, b)))})
        cases.append({"a": hx(g2_to_bytes(a)), "b": hx(g2_to_bytes(g2_neg(a))),
                      "add": hx(g2_to_bytes(None))})
    muls = []
    for k in [0, 1, 2, 3, R - 1, R - 2, (1 << 253) - 1, rng.next_fr()]:
        muls.append({"p": hx(g2_to_bytes(G)), "k": hx(to_canon_bytes(k % R)),
                     "mul": hx(g2_to_bytes(g2_mul(G, k)))})
    msms = []
    Q = g2_mul(G, 0xC0FFEE)
    for scalars, points in [
        ([], []),
        ([rng.next_fr()], [G]),
        ([rng.next_fr(), rng.next_fr()], [G, Q]),      # the SRS shape (n=2)
        ([0, rng.next_fr()], [G, Q]),
        ([R - 1, 1], [G, G]),                          # cancellation
    ]:
        msms.append({
            "n": len(scalars),
            "scalars_canon": [hx(to_canon_bytes(s)) for s in scalars],
            "bases": [hx(g2_to_bytes(p)) for p in points],
            "result": hx(g2_to_bytes(g2_msm(scalars, points))),
        })
    return {"add_cases": cases, "mul_cases": muls, "msm_cases": msms}


def main():
    fixtures = {
        "fr_arith.json": gen_field(R, "fr"),
        "fq_arith.json": gen_field(P, "fq"),
        "g1.json": gen_g1(),
        "g2.json": gen_g2(),
        "msm.json": gen_msm(),
        "ntt.json": gen_ntt(),
    }
    for name, data in fixtures.items():
        path = os.path.join(HERE, name)
        with open(path, "w") as f:
            json.dump(data, f, indent=1)
        print(f"wrote {path} ({os.path.getsize(path)} bytes)")


if __name__ == "__main__":
    main()
