"""GPU parity tests (@gpu): the HIP path vs the CPU oracle and the committed
golden vectors, bit-exact, including the BASELINE-size property checks."""
import hashlib

import pytest

pytestmark = pytest.mark.gpu


def hx(s):
    return bytes.fromhex(s)


def _omega(log_n, oracle, golden):
    for c in golden("ntt.json")["seeded_cases"]:
        if c["log_n"] == 12:
            w = hx(c["omega_mont"])
    if log_n > 12:  # lift via the 2^28 root: omega_28^(2^(28-log_n))
        w = ROOT28
        for _ in range(28 - log_n):
            w = oracle.fr_mul(w, w)
    else:
        for _ in range(12 - log_n):
            w = oracle.fr_mul(w, w)
    return w


# 2^28 primitive root of BN254 Fr, Montgomery form (= 7^((r-1)/2^28));
# verified against the committed 2^12 fixture root in test_root_consistency.
ROOT28 = None


@pytest.fixture(scope="module", autouse=True)
def _root28(oracle):
    global ROOT28
    # GENERATOR=7 canonical -> mont; exponent (r-1)/2^28
    r = 21888242871839275222246405745257275088548364400416034343698204186575808495617
    g7 = oracle.fr_from_canonical((7).to_bytes(32, "little"))
    e = ((r - 1) >> 28).to_bytes(32, "little")
    ROOT28 = oracle.fr_pow(g7, e)
    yield


def test_root_consistency(oracle, golden):
    w = ROOT28
    for _ in range(28 - 12):
        w = oracle.fr_mul(w, w)
    for c in golden("ntt.json")["seeded_cases"]:
        if c["log_n"] == 12:
            assert w.hex() == c["omega_mont"]


# ---------------------------------------------------------------- MSM
def test_msm_inline_golden(gpu, golden):
    for i, c in enumerate(golden("msm.json")["inline_cases"]):
        n = c["n"]
        if n == 0:
            continue  # host-pointer path handles n=0 below
        sc = b"".join(hx(s) for s in c["scalars_canon"])
        scm = b"".join(hx(s) for s in c["scalars_mont"])
        bs = b"".join(hx(s) for s in c["bases"])
        assert gpu.msm(bs, sc, n, canonical=True).hex() == c["result"], i
        assert gpu.msm(bs, scm, n, canonical=False).hex() == c["result"], i


def test_msm_n0(gpu):
    assert gpu.msm(b"", b"", 0) == bytes(64)


def test_msm_seeded_golden(gpu, oracle, golden):
    for c in golden("msm.json")["seeded_cases"]:
        sc, bs = oracle.gen_msm_inputs(c["n"], c["seed"])
        assert gpu.msm(bs, sc, c["n"]).hex() == c["result"], c["n"]


@pytest.mark.parametrize("log_n", [12, 16])
def test_msm_vs_oracle(gpu, oracle, log_n):
    n = 1 << log_n
    sc, bs = oracle.gen_msm_inputs(n, 9000 + log_n, fast=True)
    want = oracle.msm(bs, sc, n)
    assert gpu.msm(bs, sc, n) == want


def test_msm_2pow20_vs_oracle(gpu, oracle):
    """Full BASELINE config[1] size, exact parity."""
    n = 1 << 20
    sc, bs = oracle.gen_msm_inputs(n, 42, fast=True)
    want = oracle.msm(bs, sc, n)
    assert gpu.msm(bs, sc, n) == want


@pytest.mark.parametrize("log_n", [22, 23])
def test_msm_aggregation_sizes_vs_oracle(gpu, oracle, log_n):
    """Aggregation-circuit MSM sizes (K=23/24 proofs commit at 2^23/2^24;
    2^22 covers the k=20 extended-domain h commits)."""
    n = 1 << log_n
    sc, bs = oracle.gen_msm_inputs(n, 2200 + log_n, fast=True)
    want = oracle.msm(bs, sc, n)
    assert gpu.msm(bs, sc, n) == want


def test_msm_bases_cache(gpu, oracle):
    n = 4096
    sc, bs = oracle.gen_msm_inputs(n, 31, fast=True)
    r1 = gpu.msm(bs, sc, n, bases_id=77)
    r2 = gpu.msm(None, sc, n, bases_id=77)  # cached upload
    assert r1 == r2 == oracle.msm(bs, sc, n)


def test_msm_async_pipeline_matches_sync(gpu, oracle):
    """Depth-2 pipelined MSMs on the two per-device slots: every in-flight
    call returns exactly the synchronous result, including when two
    different inputs are in flight on the two slots at once."""
    from spectre_amd import ffi
    n = 1 << 14
    sc_a, bs = oracle.gen_msm_inputs(n, 61, fast=True)
    sc_b = oracle.gen_msm_inputs(n, 62, fast=True)[0]
    d_b = gpu.malloc(64 * n)
    d_sa = gpu.malloc(32 * n)
    d_sb = gpu.malloc(32 * n)
    gpu.upload(d_b, bs)
    gpu.upload(d_sa, sc_a)
    gpu.upload(d_sb, sc_b)
    want_a = ffi.combine_partials(gpu.msm_shard_device(d_b, d_sa, n), 1)
    want_b = ffi.combine_partials(gpu.msm_shard_device(d_b, d_sb, n), 1)
    # interleave: a,b,a,b,... with both slots busy simultaneously
    pend = []
    got = []
    for i in range(6):
        buf, slot = gpu.msm_shard_device_async(d_b, d_sa if i % 2 == 0 else d_sb, n)
        pend.append((buf, slot))
        if len(pend) == 2:
            b0, s0 = pend.pop(0)
            gpu.msm_slot_wait(s0)
            got.append(ffi.combine_partials(bytes(b0), 1))
    for b0, s0 in pend:
        gpu.msm_slot_wait(s0)
        got.append(ffi.combine_partials(bytes(b0), 1))
    for i, g in enumerate(got):
        assert g == (want_a if i % 2 == 0 else want_b), f"call {i}"
    for p in (d_b, d_sa, d_sb):
        gpu.free(p)


def test_msm_two_device_in_library(oracle):
    """In-library multi-device sharding (spectre_gpu_msm_g1 num_gpus=2) must
    equal the single-device result bit-for-bit. Skips on 1-GPU boxes (the
    per-round driver box); armed for the 8-GPU scaling node (VERDICT r01
    item 9: keep the 8-GPU path warm)."""
    import torch
    if torch.cuda.device_count() < 2:
        pytest.skip("needs >= 2 devices")
    from spectre_amd import SpectreGpu
    g = SpectreGpu([0, 1])
    try:
        n = 1 << 14
        sc, bs = oracle.gen_msm_inputs(n, 63, fast=True)
        one = g.msm(bs, sc, n, num_gpus=1)
        two = g.msm(bs, sc, n, num_gpus=2)
        assert one == two
        assert one == oracle.msm(bs, sc, n)
    finally:
        g.close()


def test_msm_window_shard_combine_matches_direct(gpu, oracle):
    """Window-sharded shards (disjoint window ranges over ALL points,
    concatenated in rank order) must equal the direct result bit-for-bit —
    the default N>1 exchange (pure allgather, no reduction)."""
    from spectre_amd import ffi
    n = 1 << 13
    sc, bs = oracle.gen_msm_inputs(n, 64, fast=True)
    direct = gpu.msm(bs, sc, n)
    d_b = gpu.malloc(64 * n)
    d_s = gpu.malloc(32 * n)
    gpu.upload(d_b, bs)
    gpu.upload(d_s, sc)
    for world in (2, 4, 8):
        w_cnt = ffi.NUM_WINDOWS // world
        blob = b"".join(
            gpu.msm_shard_windows_device(d_b, d_s, n, r * w_cnt, w_cnt)
            for r in range(world))
        assert ffi.combine_window_partials(blob, world) == direct, world
    # async variant, interleaved across two ranks' window sets
    w_cnt = ffi.NUM_WINDOWS // 2
    b0, s0 = gpu.msm_shard_windows_device_async(d_b, d_s, n, 0, w_cnt)
    b1, s1 = gpu.msm_shard_windows_device_async(d_b, d_s, n, w_cnt, w_cnt)
    gpu.msm_slot_wait(s0)
    gpu.msm_slot_wait(s1)
    assert ffi.combine_window_partials(bytes(b0) + bytes(b1), 2) == direct
    gpu.free(d_b)
    gpu.free(d_s)


def test_msm_shard_combine_matches_direct(gpu, oracle):
    """Two shards on one device + host combine == unsharded result — the
    exact exchange the multi-GPU path performs (bit-identical by affine
    canonicality)."""
    from spectre_amd import ffi
    n = 1 << 13
    sc, bs = oracle.gen_msm_inputs(n, 60, fast=True)
    direct = gpu.msm(bs, sc, n)
    half = n // 2
    parts = b""
    d_b = gpu.malloc(64 * half)
    d_s = gpu.malloc(32 * half)
    for lo in (0, half):
        gpu.upload(d_b, bs[64 * lo:64 * (lo + half)])
        gpu.upload(d_s, sc[32 * lo:32 * (lo + half)])
        parts += gpu.msm_shard_device(d_b, d_s, half)
    gpu.free(d_b)
    gpu.free(d_s)
    assert ffi.combine_partials(parts, 2) == direct


def test_msm_batch_matches_individual(gpu, oracle):
    """Batched MSM (shared bases, fused pipeline) must equal per-vector
    results bit-for-bit."""
    n, nbatch = 2048, 5
    _, bs = oracle.gen_msm_inputs(n, 70, fast=True)
    scal = b""
    singles = []
    for b in range(nbatch):
        sc, _ = oracle.gen_msm_inputs(n, 71 + b, fast=True)
        scal += sc
        singles.append(oracle.msm(bs, sc, n))
    got = gpu.msm_batch(bs, scal, nbatch, n)
    assert got == singles
    # edge: batch with an all-zero vector and a duplicate vector
    scal2 = bytes(32 * n) + scal[:32 * n] + scal[:32 * n]
    got2 = gpu.msm_batch(bs, scal2, 3, n)
    assert got2[0] == bytes(64)
    assert got2[1] == got2[2] == singles[0]


def test_msm_device_resident(gpu, oracle):
    n = 4096
    sc, bs = oracle.gen_msm_inputs(n, 61, fast=True)
    d_b = gpu.malloc(64 * n)
    d_s = gpu.malloc(32 * n)
    gpu.upload(d_b, bs)
    gpu.upload(d_s, sc)
    got = gpu.msm_device(d_b, d_s, n)
    gpu.free(d_b)
    gpu.free(d_s)
    assert got == oracle.msm(bs, sc, n)


# ---------------------------------------------------------------- NTT
def test_ntt_inline_golden(gpu, golden):
    for c in golden("ntt.json")["inline_cases"]:
        ln = c["log_n"]
        inp = b"".join(hx(x) for x in c["input_mont"])
        om = hx(c["omega_mont"])
        assert gpu.ntt(inp, ln, om).hex() == "".join(c["ntt"]), (ln, "fwd")


def test_ntt_inline_golden_inverse_coset(gpu, oracle, golden):
    for c in golden("ntt.json")["inline_cases"]:
        ln = c["log_n"]
        inp = b"".join(hx(x) for x in c["input_mont"])
        om = hx(c["omega_mont"])
        out_i = gpu.ntt(inp, ln, oracle.fr_inv(om), inverse=True)
        assert out_i.hex() == "".join(c["intt"]), (ln, "inv")
        out_c = gpu.ntt(inp, ln, om, coset_gen=hx(c["coset_g_mont"]))
        assert out_c.hex() == "".join(c["coset_ntt"]), (ln, "coset")


def test_ntt_seeded_golden(gpu, oracle, golden):
    for c in golden("ntt.json")["seeded_cases"]:
        ln = c["log_n"]
        inp = oracle.gen_fr_vector(1 << ln, c["seed"])
        om = hx(c["omega_mont"])
        assert hashlib.sha256(gpu.ntt(inp, ln, om)).hexdigest() == c["ntt_sha256"]
        out_i = gpu.ntt(inp, ln, oracle.fr_inv(om), inverse=True)
        assert hashlib.sha256(out_i).hexdigest() == c["intt_sha256"]


@pytest.mark.parametrize("log_n", [13, 16, 18])
def test_ntt_vs_oracle(gpu, oracle, golden, log_n):
    """Two-pass GPU path vs oracle at sizes the oracle runs in seconds."""
    n = 1 << log_n
    a = oracle.gen_fr_vector(n, 100 + log_n)
    w = _omega(log_n, oracle, golden)
    assert gpu.ntt(a, log_n, w) == oracle.ntt(a, log_n, w)
    wi = oracle.fr_inv(w)
    assert gpu.ntt(a, log_n, wi, inverse=True) == oracle.ntt(a, log_n, wi,
                                                             inverse=True)


def test_ntt_coset_vs_oracle(gpu, oracle, golden):
    log_n = 14
    n = 1 << log_n
    a = oracle.gen_fr_vector(n, 777)
    w = _omega(log_n, oracle, golden)
    g = oracle.fr_from_canonical((5).to_bytes(32, "little"))
    assert gpu.ntt(a, log_n, w, coset_gen=g) == oracle.ntt(a, log_n, w,
                                                           coset_gen=g)
    gi = oracle.fr_inv(g)
    wi = oracle.fr_inv(w)
    assert (gpu.ntt(a, log_n, wi, inverse=True, coset_gen=gi)
            == oracle.ntt(a, log_n, wi, inverse=True, coset_gen=gi))


@pytest.mark.parametrize("log_n", [20, 23])
def test_ntt_roundtrip_full_size(gpu, oracle, golden, log_n):
    """BASELINE-size property check (oracle-free at this size):
    iNTT(NTT(a)) == a and coset round trip, device-resident."""
    n = 1 << log_n
    a = oracle.gen_fr_vector(n, 4000 + log_n)
    w = _omega(log_n, oracle, golden)
    wi = oracle.fr_inv(w)
    d = gpu.malloc(32 * n)
    gpu.upload(d, a)
    gpu.ntt_device(d, log_n, w)
    gpu.ntt_device(d, log_n, wi, inverse=True)
    assert gpu.download(d, 32 * n) == a
    g = oracle.fr_from_canonical((5).to_bytes(32, "little"))
    gpu.ntt_device(d, log_n, w, coset_gen=g)
    gpu.ntt_device(d, log_n, wi, inverse=True, coset_gen=oracle.fr_inv(g))
    assert gpu.download(d, 32 * n) == a
    gpu.free(d)


def test_ntt_full_size_spotcheck_vs_dft(gpu, oracle, golden):
    """At 2^20, check a handful of output positions against a direct
    DFT evaluation out[j] = sum_i a_i w^(ij) computed with Python bigints."""
    log_n = 20
    n = 1 << log_n
    r = 21888242871839275222246405745257275088548364400416034343698204186575808495617
    R = 1 << 256

    def from_mont(b):
        return int.from_bytes(b, "little") * pow(R, -1, r) % r

    a = oracle.gen_fr_vector(n, 5555)
    w = _omega(log_n, oracle, golden)
    out = gpu.ntt(a, log_n, w)
    wv = from_mont(w)
    av = [from_mont(a[32 * i:32 * i + 32]) for i in range(n)]
    import random
    rng = random.Random(1)
    for j in [0, 1, n - 1] + [rng.randrange(n) for _ in range(2)]:
        wj = pow(wv, j, r)
        acc, cur = 0, 1
        for i in range(n):
            acc = (acc + av[i] * cur) % r
            cur = cur * wj % r
        got = from_mont(out[32 * j:32 * j + 32])
        assert got == acc, f"output {j}"


# ---------------------------------------------------------------- robustness
def test_concurrent_calls_one_ctx(gpu, oracle):
    """halo2 commits from concurrent rayon workers; the ctx must serialize
    safely and return correct results from every thread."""
    import threading
    n = 1024
    cases = []
    for seed in range(6):
        sc, bs = oracle.gen_msm_inputs(n, 800 + seed, fast=True)
        cases.append((sc, bs, oracle.msm(bs, sc, n)))
    errors = []

    def worker(sc, bs, want):
        try:
            for _ in range(3):
                got = gpu.msm(bs, sc, n)
                assert got == want
        except Exception as e:  # pragma: no cover
            errors.append(e)

    threads = [threading.Thread(target=worker, args=c) for c in cases]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errors


def test_error_paths(gpu):
    import pytest as _pytest
    with _pytest.raises(RuntimeError, match="log_n"):
        gpu.ntt(b"\x00" * 32, 29, b"\x01" + b"\x00" * 31)
    with _pytest.raises(RuntimeError, match="nbatch"):
        gpu.msm_batch(b"\x00" * 64, b"\x00" * (33 * 32), 33, 1)


def test_ntt_three_pass_forced_vs_oracle(oracle, golden):
    """The 3-pass (log_n>24) path, forced at oracle-checkable sizes via
    SPECTRE_NTT_FORCE3, must match the oracle bit-exactly (fwd/inv/coset)."""
    import os
    import subprocess
    import sys
    REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    code = r'''
import sys, os
sys.path.insert(0, %r); sys.path.insert(0, %r)
import pywrap as oracle
from spectre_amd import SpectreGpu
import json
fix = json.load(open(os.path.join(%r, "tests", "golden", "ntt.json")))
w12 = bytes.fromhex([c for c in fix["seeded_cases"] if c["log_n"] == 12][0]["omega_mont"])
gpu = SpectreGpu([0])
g = oracle.fr_from_canonical((5).to_bytes(32, "little"))
for log_n in (6, 10, 12, 14):
    w = w12
    if log_n <= 12:
        for _ in range(12 - log_n):
            w = oracle.fr_mul(w, w)
    else:
        r = 21888242871839275222246405745257275088548364400416034343698204186575808495617
        g7 = oracle.fr_from_canonical((7).to_bytes(32, "little"))
        w = oracle.fr_pow(g7, ((r - 1) >> 28).to_bytes(32, "little"))
        for _ in range(28 - log_n):
            w = oracle.fr_mul(w, w)
    a = oracle.gen_fr_vector(1 << log_n, 600 + log_n)
    assert gpu.ntt(a, log_n, w) == oracle.ntt(a, log_n, w), log_n
    wi = oracle.fr_inv(w)
    assert gpu.ntt(a, log_n, wi, inverse=True) == oracle.ntt(a, log_n, wi, inverse=True), log_n
    assert gpu.ntt(a, log_n, w, coset_gen=g) == oracle.ntt(a, log_n, w, coset_gen=g), log_n
print("force3 ok")
''' % (REPO, REPO + "/oracle", REPO)
    env = dict(os.environ, SPECTRE_NTT_FORCE3="1")
    r = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    assert "force3 ok" in r.stdout, r.stdout + r.stderr


def test_ntt_2pow25_roundtrip(gpu, oracle, golden):
    """Aggregation-scale extended domain (>2^24): 3-pass path round trip
    and parity vs the oracle on a slice via linear DFT check is too slow;
    round trip + oracle comparison at full size (OpenMP oracle, ~seconds)."""
    log_n = 25
    n = 1 << log_n
    r = 21888242871839275222246405745257275088548364400416034343698204186575808495617
    g7 = oracle.fr_from_canonical((7).to_bytes(32, "little"))
    w = oracle.fr_pow(g7, ((r - 1) >> 28).to_bytes(32, "little"))
    for _ in range(28 - log_n):
        w = oracle.fr_mul(w, w)
    wi = oracle.fr_inv(w)
    chunk = oracle.gen_fr_vector(1 << 20, 2525)
    a = chunk * (n >> 20)
    d = gpu.malloc(32 * n)
    gpu.upload(d, a)
    gpu.ntt_device(d, log_n, w)
    out_gpu_head = gpu.download(d, 32 * 4096)  # spot region
    want = oracle.ntt(a, log_n, w)
    assert out_gpu_head == want[:32 * 4096]
    assert gpu.download(d, 32 * n) == want
    gpu.ntt_device(d, log_n, wi, inverse=True)
    assert gpu.download(d, 32 * n) == a
    gpu.free(d)


def _ref_gate_eval(cols_int, consts_int, program, n, rot_scale, r_mod,
                   y_int=None, prev=None):
    """Bigint restatement of the gate-expression evaluator semantics
    (the checker for SURVEY §8f-3): stack machine per row, rotations
    (row + rot*rot_scale) mod n, out = prev*y + v when y given."""
    out = []
    for row in range(n):
        st = []
        for op, a, b in program:
            if op == 0:
                st.append(cols_int[a][(row + b * rot_scale) % n])
            elif op == 1:
                st.append(consts_int[a])
            elif op == 5:
                st[-1] = (-st[-1]) % r_mod
            else:
                rhs = st.pop()
                lhs = st.pop()
                st.append((lhs + rhs) % r_mod if op == 2 else
                          (lhs - rhs) % r_mod if op == 3 else
                          lhs * rhs % r_mod)
        v = st[0]
        if y_int is not None:
            v = (prev[row] * y_int + v) % r_mod
        out.append(v)
    return out


def test_gate_eval_vs_reference(gpu, oracle):
    """Gate-expression evaluator (quotient phase): the
    halo2-base flex-gate shape q*(a + b*c - d) over rotations of one
    column, plus random programs, vs the bigint restatement."""
    import random
    R = 21888242871839275222246405745257275088548364400416034343698204186575808495617
    n = 1 << 10
    rng = random.Random(8015)

    def fr_vec(ints):
        return b"".join(
            oracle.fr_from_canonical(v.to_bytes(32, "little")) for v in ints)

    def to_ints(raw):
        return [int.from_bytes(
            oracle.fr_to_canonical(raw[32 * i:32 * (i + 1)]), "little")
            for i in range(len(raw) // 32)]

    ncols = 3
    cols_int = [[rng.randrange(R) for _ in range(n)] for _ in range(ncols)]
    d_cols = []
    for ci in cols_int:
        d = gpu.malloc(32 * n)
        gpu.upload(d, fr_vec(ci))
        d_cols.append(d)
    d_out = gpu.malloc(32 * n)

    G = gpu
    # flex gate: q * (a + b*c - d) with a,b,c,d = rotations 0..3 of col 1,
    # q = col 0 (rot_scale 4 as the extended domain would use)
    flex = [(G.GATE_COL, 0, 0),
            (G.GATE_COL, 1, 0), (G.GATE_COL, 1, 1), (G.GATE_COL, 1, 2),
            (G.GATE_MUL, 0, 0), (G.GATE_ADD, 0, 0),
            (G.GATE_COL, 1, 3), (G.GATE_SUB, 0, 0),
            (G.GATE_MUL, 0, 0)]
    G.gate_eval(d_cols, b"", flex, n, rot_scale=4, d_out=d_out)
    want = _ref_gate_eval(cols_int, [], flex, n, 4, R)
    assert to_ints(bytes(G.download(d_out, 32 * n))) == want

    # accumulate form out = out*y + v, with constants and negation
    y = rng.randrange(R)
    consts = [rng.randrange(R) for _ in range(2)]
    prog2 = [(G.GATE_CONST, 0, 0), (G.GATE_COL, 2, -1), (G.GATE_MUL, 0, 0),
             (G.GATE_CONST, 1, 0), (G.GATE_NEG, 0, 0), (G.GATE_ADD, 0, 0)]
    G.gate_eval(d_cols, fr_vec(consts), prog2, n, rot_scale=1,
                y=fr_vec([y]), d_out=d_out)
    want2 = _ref_gate_eval(cols_int, consts, prog2, n, 1, R, y, want)
    assert to_ints(bytes(G.download(d_out, 32 * n))) == want2

    # random well-formed programs to max depth
    for trial in range(6):
        prog, depth, maxd = [], 0, 0
        consts = [rng.randrange(R) for _ in range(3)]
        while len(prog) < 20 or depth != 1:
            if depth >= 2 and (depth >= 7 or rng.random() < 0.5):
                prog.append((rng.choice([G.GATE_ADD, G.GATE_SUB, G.GATE_MUL]),
                             0, 0))
                depth -= 1
            elif depth >= 1 and rng.random() < 0.1:
                prog.append((G.GATE_NEG, 0, 0))
            elif rng.random() < 0.7:
                prog.append((G.GATE_COL, rng.randrange(ncols),
                             rng.randrange(-3, 4)))
                depth += 1
            else:
                prog.append((G.GATE_CONST, rng.randrange(3), 0))
                depth += 1
            maxd = max(maxd, depth)
        G.gate_eval(d_cols, fr_vec(consts), prog, n, rot_scale=2, d_out=d_out)
        want = _ref_gate_eval(cols_int, consts, prog, n, 2, R)
        assert to_ints(bytes(G.download(d_out, 32 * n))) == want, trial

    # validation: malformed programs are rejected loudly
    import pytest as _pytest
    for bad in ([(G.GATE_ADD, 0, 0)],                     # underflow
                [(G.GATE_COL, 99, 0)],                    # bad column
                [(G.GATE_COL, 0, 0), (G.GATE_COL, 0, 0)],  # 2 left on stack
                [(G.GATE_COL, 0, 0)] * 9):                # depth > max
        with _pytest.raises(RuntimeError):
            G.gate_eval(d_cols, b"", bad, n, d_out=d_out)

    for d in d_cols + [d_out]:
        gpu.free(d)


def test_quotient_pipeline_device_resident(gpu, oracle):
    """The whole quotient phase device-resident: per-column
    iFFT -> coset-FFT, gate expression over the coset values, inverse-coset
    iFFT of the result — one upload, one download, bit-exact against the
    oracle NTTs + a bigint evaluation of the same expression."""
    R = 21888242871839275222246405745257275088548364400416034343698204186575808495617
    log_n = 10
    n = 1 << log_n
    # omega from the module ROOT28 (verified vs fixtures in test_root_consistency)
    w = ROOT28
    for _ in range(28 - log_n):
        w = oracle.fr_mul(w, w)
    wi = oracle.fr_inv(w)
    g5 = oracle.fr_from_canonical((5).to_bytes(32, "little"))
    g5i = oracle.fr_inv(g5)

    cols = [oracle.gen_fr_vector(n, 7000 + i) for i in range(3)]
    d_cols = []
    for cvec in cols:
        d = gpu.malloc(32 * n)
        gpu.upload(d, cvec)
        d_cols.append(d)
    d_out = gpu.malloc(32 * n)

    # device: evaluations -> coeffs -> coset evals, per column
    for d in d_cols:
        gpu.ntt_device(d, log_n, wi, inverse=True)
        gpu.ntt_device(d, log_n, w, coset_gen=g5)
    G = gpu
    flex = [(G.GATE_COL, 0, 0),
            (G.GATE_COL, 1, 0), (G.GATE_COL, 1, 1), (G.GATE_COL, 2, 2),
            (G.GATE_MUL, 0, 0), (G.GATE_ADD, 0, 0),
            (G.GATE_COL, 2, 3), (G.GATE_SUB, 0, 0),
            (G.GATE_MUL, 0, 0)]
    G.gate_eval(d_cols, b"", flex, n, rot_scale=4, d_out=d_out)
    gpu.ntt_device(d_out, log_n, wi, inverse=True, coset_gen=g5i)
    got = bytes(gpu.download(d_out, 32 * n))

    # oracle: identical sequence
    ref_cosets = []
    for cvec in cols:
        coeff = oracle.ntt(cvec, log_n, wi, inverse=True)
        ref_cosets.append(oracle.ntt(coeff, log_n, w, coset_gen=g5))

    def to_ints(raw):
        return [int.from_bytes(
            oracle.fr_to_canonical(raw[32 * i:32 * (i + 1)]), "little")
            for i in range(n)]
    ci = [to_ints(c) for c in ref_cosets]
    ev = _ref_gate_eval(ci, [], flex, n, 4, R)
    ev_bytes = b"".join(
        oracle.fr_from_canonical(v.to_bytes(32, "little")) for v in ev)
    want = oracle.ntt(ev_bytes, log_n, wi, inverse=True, coset_gen=g5i)
    assert got == want
    for d in d_cols + [d_out]:
        gpu.free(d)


def test_fr_vec_ops_vs_oracle(gpu, oracle):
    """Pointwise Fr vector ops (quotient gate-eval glue) vs the oracle."""
    n = 4097
    a = oracle.gen_fr_vector(n, 91)
    b = oracle.gen_fr_vector(n, 92)
    cst = oracle.gen_fr_vector(1, 93)
    d_a = gpu.malloc(32 * n)
    d_b = gpu.malloc(32 * n)
    d_o = gpu.malloc(32 * n)
    gpu.upload(d_a, a)
    gpu.upload(d_b, b)
    cases = [
        (gpu.VEC_ADD, None, oracle.fr_add),
        (gpu.VEC_SUB, None, oracle.fr_sub),
        (gpu.VEC_MUL, None, oracle.fr_mul),
    ]
    for op, c, fn in cases:
        gpu.fr_vec_op(op, d_a, d_b, c, d_o, n)
        got = gpu.download(d_o, 32 * n)
        want = b"".join(fn(a[32 * i:32 * i + 32], b[32 * i:32 * i + 32])
                        for i in range(n))
        assert got == want, op
    gpu.fr_vec_op(gpu.VEC_SCALE, d_a, None, cst, d_o, n)
    got = gpu.download(d_o, 32 * n)
    want = b"".join(oracle.fr_mul(a[32 * i:32 * i + 32], cst)
                    for i in range(n))
    assert got == want
    gpu.fr_vec_op(gpu.VEC_ADD_SCALED, d_a, d_b, cst, d_o, n)
    got = gpu.download(d_o, 32 * n)
    want = b"".join(
        oracle.fr_add(a[32 * i:32 * i + 32],
                      oracle.fr_mul(b[32 * i:32 * i + 32], cst))
        for i in range(n))
    assert got == want
    # in-place aliasing (out == a)
    gpu.fr_vec_op(gpu.VEC_ADD, d_a, d_b, None, d_a, n)
    got = gpu.download(d_a, 32 * n)
    want = b"".join(oracle.fr_add(a[32 * i:32 * i + 32], b[32 * i:32 * i + 32])
                    for i in range(n))
    assert got == want
    for p in (d_a, d_b, d_o):
        gpu.free(p)


def test_msm_randomized_sizes(gpu, oracle):
    """Seeded sweep over awkward sizes (odd, prime-ish, off-by-one around
    thread/partition boundaries) x scalar encodings — edge insurance for the
    equal-work partitioning and boundary-fixup logic."""
    import random
    rng = random.Random(20260915)
    sizes = [1, 2, 63, 64, 65, 255, 1000003 % 4096, 4095, 4097,
             rng.randrange(1, 20000), rng.randrange(1, 20000)]
    for idx, n in enumerate(sizes):
        sc, bs = oracle.gen_msm_inputs(n, 7000 + idx, fast=True)
        want = oracle.msm(bs, sc, n)
        assert gpu.msm(bs, sc, n) == want, (n, "canonical")
        scm = b"".join(
            oracle.fr_from_canonical(sc[32 * i:32 * i + 32])
            for i in range(n))
        assert gpu.msm(bs, scm, n, canonical=False) == want, (n, "mont")


def test_msm_batch_randomized(gpu, oracle):
    import random
    rng = random.Random(99)
    for trial in range(3):
        n = rng.randrange(100, 5000)
        nb = rng.randrange(2, 7)
        _, bs = oracle.gen_msm_inputs(n, 7100 + trial, fast=True)
        scal, singles = b"", []
        for b in range(nb):
            sc, _ = oracle.gen_msm_inputs(n, 7200 + 10 * trial + b, fast=True)
            scal += sc
            singles.append(oracle.msm(bs, sc, n))
        assert gpu.msm_batch(bs, scal, nb, n) == singles, (n, nb)
