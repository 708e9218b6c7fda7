"""Size-independent algebraic properties of the oracle (these same
properties re-run on the GPU path at full BASELINE sizes in test_gpu_parity)."""
import json
import os

import pytest

HERE = os.path.dirname(os.path.abspath(__file__))


def _omega(log_n, oracle, golden):
    """Derive a 2^log_n root of unity by squaring down the committed 2^12 root."""
    for c in golden("ntt.json")["seeded_cases"]:
        if c["log_n"] == 12:
            w = bytes.fromhex(c["omega_mont"])
    for _ in range(12 - log_n):
        w = oracle.fr_mul(w, w)
    return w


def test_ntt_roundtrip(oracle, golden):
    for log_n in [4, 8, 10, 12]:
        n = 1 << log_n
        a = oracle.gen_fr_vector(n, 99 + log_n)
        w = _omega(log_n, oracle, golden)
        fwd = oracle.ntt(a, log_n, w)
        back = oracle.ntt(fwd, log_n, oracle.fr_inv(w), inverse=True)
        assert back == a


def test_ntt_coset_roundtrip(oracle, golden):
    log_n = 10
    a = oracle.gen_fr_vector(1 << log_n, 4242)
    w = _omega(log_n, oracle, golden)
    g = oracle.fr_from_canonical((5).to_bytes(32, "little"))
    fwd = oracle.ntt(a, log_n, w, coset_gen=g)
    back = oracle.ntt(fwd, log_n, oracle.fr_inv(w), inverse=True,
                      coset_gen=oracle.fr_inv(g))
    assert back == a


def test_ntt_linearity(oracle, golden):
    log_n = 8
    n = 1 << log_n
    w = _omega(log_n, oracle, golden)
    a = oracle.gen_fr_vector(n, 1)
    b = oracle.gen_fr_vector(n, 2)
    apb = b"".join(
        oracle.fr_add(a[32 * i:32 * i + 32], b[32 * i:32 * i + 32])
        for i in range(n))
    fa, fb, fab = (oracle.ntt(x, log_n, w) for x in (a, b, apb))
    want = b"".join(
        oracle.fr_add(fa[32 * i:32 * i + 32], fb[32 * i:32 * i + 32])
        for i in range(n))
    assert fab == want


def test_msm_matches_bruteforce(oracle):
    n = 20
    sc, bs = oracle.gen_msm_inputs(n, 555)
    acc = bytes(64)
    for i in range(n):
        term = oracle.g1_mul(bs[64 * i:64 * i + 64], sc[32 * i:32 * i + 32])
        acc = oracle.g1_add(acc, term)
    assert oracle.msm(bs, sc, n) == acc


def test_msm_linearity(oracle):
    """MSM(a, P) + MSM(b, P) == MSM(a+b mod r, P)."""
    n = 64
    sa, bs = oracle.gen_msm_inputs(n, 10)
    sb, _ = oracle.gen_msm_inputs(n, 20)
    r = 21888242871839275222246405745257275088548364400416034343698204186575808495617
    sab = b"".join(
        ((int.from_bytes(sa[32 * i:32 * i + 32], "little")
          + int.from_bytes(sb[32 * i:32 * i + 32], "little")) % r
         ).to_bytes(32, "little") for i in range(n))
    lhs = oracle.g1_add(oracle.msm(bs, sa, n), oracle.msm(bs, sb, n))
    assert lhs == oracle.msm(bs, sab, n)


def test_msm_fast_gen_on_curve(oracle):
    _, bs = oracle.gen_msm_inputs(64, 30, fast=True)
    for i in range(64):
        assert oracle.g1_is_on_curve(bs[64 * i:64 * i + 64])


def test_g2_group_properties(oracle):
    """Randomized G2 algebra: commutativity, associativity, doubling
    consistency, scalar-mul additivity, negation — independent of the
    committed fixtures (which pin values; these pin structure)."""
    import random
    rng = random.Random(90210)
    R = 21888242871839275222246405745257275088548364400416034343698204186575808495617
    import json
    import os
    g2fix = json.load(open(os.path.join(
        os.path.dirname(__file__), "golden", "g2.json")))
    G = bytes.fromhex(g2fix["mul_cases"][1]["mul"])  # 1*G = generator
    assert oracle.g2_is_on_curve(G)
    for _ in range(6):
        a, b = rng.randrange(R), rng.randrange(R)
        P = oracle.g2_mul(G, a.to_bytes(32, "little"))
        Q = oracle.g2_mul(G, b.to_bytes(32, "little"))
        assert oracle.g2_is_on_curve(P) and oracle.g2_is_on_curve(Q)
        # commutativity
        assert oracle.g2_add(P, Q) == oracle.g2_add(Q, P)
        # additivity: aG + bG == (a+b)G
        S = oracle.g2_mul(G, ((a + b) % R).to_bytes(32, "little"))
        assert oracle.g2_add(P, Q) == S
        # doubling == adding to itself == 2k*G
        D = oracle.g2_add(P, P)
        assert D == oracle.g2_mul(G, (2 * a % R).to_bytes(32, "little"))
        # negation cancels
        assert oracle.g2_add(P, oracle.g2_neg(P)) == bytes(128)
        # associativity with a third point
        T = oracle.g2_mul(G, rng.randrange(R).to_bytes(32, "little"))
        assert (oracle.g2_add(oracle.g2_add(P, Q), T)
                == oracle.g2_add(P, oracle.g2_add(Q, T)))
