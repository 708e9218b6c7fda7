"""Pure-Python (bigint) BN254 reference used ONLY to generate golden vectors.

TEST INFRASTRUCTURE — this module is an independent restatement of the
algorithms the reference's hot path computes; it must never be imported by
product code. It exists so that the C oracle (oracle/) and the HIP kernels
can both be pinned against a third, independent implementation whose
arithmetic engine (Python bigints) is trusted.

Reference algorithms restated (reference = ChainSafe/Spectre at
/root/reference; the arithmetic itself lives in its un-vendored third-party
dependency `halo2curves-axiom = 0.5.2` (Cargo.toml:53) and the PSE
`halo2_proofs` fork, called from lightclient-circuits/src/util/circuit.rs:158,177,211):

  * BN254 (alt_bn128) base field Fq, scalar field Fr, G1: y^2 = x^3 + 3.
    Constants are the published alt_bn128/EIP-196 parameters.
  * `best_multiexp(coeffs, bases)`: mathematically Sum_i coeffs_i * bases_i
    over G1 — the output is algorithm-independent (any windowing gives the
    same group element), so parity is defined by the value, not the schedule.
  * `best_fft(a, omega, log_n)`: the in-place radix-2 DFT
    out[j] = Sum_i a[i] * omega^(i*j)  (omega a 2^log_n-th root of unity).
  * Memory/wire formats of halo2curves 0.5.2: a field element in memory is
    4 x u64 little-endian limbs of the Montgomery residue a*R mod m with
    R = 2^256; `Fr::to_repr()` is 32 LE bytes of the canonical value;
    G1Affine in memory is (x, y) Montgomery Fq pairs, identity = (0, 0).

No file under /root/reference is read at runtime by any test; this script is
self-contained and deterministic.
"""

# ---------------------------------------------------------------- constants
# alt_bn128 / BN254 (EIP-196/197 curve), as used by halo2curves-axiom 0.5.2.
P = 21888242871839275222246405745257275088696311157297823662689037894645226208583  # Fq modulus
R = 21888242871839275222246405745257275088548364400416034343698204186575808495617  # Fr modulus
MONT = 1 << 256  # Montgomery radix for 4x64 / 8x32 limb representations

# Fr multiplicative generator (halo2curves: GENERATOR = 7) and 2-adicity 28.
FR_GEN = 7
FR_TWO_ADICITY = 28
assert (R - 1) % (1 << FR_TWO_ADICITY) == 0

def fr_root_of_unity(log_n: int) -> int:
    """A primitive 2^log_n-th root of unity in Fr (matches halo2curves'
    ROOT_OF_UNITY construction: GENERATOR^((r-1)/2^28), then squared down)."""
    assert log_n <= FR_TWO_ADICITY
    w = pow(FR_GEN, (R - 1) >> FR_TWO_ADICITY, R)
    for _ in range(FR_TWO_ADICITY - log_n):
        w = w * w % R
    return w

# sanity: order is exactly 2^28
_w = fr_root_of_unity(FR_TWO_ADICITY)
assert pow(_w, 1 << FR_TWO_ADICITY, R) == 1 and pow(_w, 1 << (FR_TWO_ADICITY - 1), R) != 1


# ---------------------------------------------------------------- encodings
def to_mont_bytes(a: int, m: int) -> bytes:
    """Montgomery-form memory image: 32 LE bytes of a*R mod m."""
    return (a * MONT % m).to_bytes(32, "little")

def from_mont_bytes(b: bytes, m: int) -> int:
    return int.from_bytes(b, "little") * pow(MONT, -1, m) % m

def to_canon_bytes(a: int) -> bytes:
    return a.to_bytes(32, "little")

def g1_to_bytes(pt) -> bytes:
    """G1Affine memory image: x||y Montgomery Fq, identity = 64 zero bytes."""
    if pt is None:
        return bytes(64)
    x, y = pt
    return to_mont_bytes(x, P) + to_mont_bytes(y, P)

def g1_from_bytes(b: bytes):
    x = from_mont_bytes(b[:32], P)
    y = from_mont_bytes(b[32:], P)
    if x == 0 and y == 0:
        return None
    return (x, y)


# ---------------------------------------------------------------- G1 affine
G1_GEN = (1, 2)

def g1_is_on_curve(pt) -> bool:
    if pt is None:
        return True
    x, y = pt
    return (y * y - x * x * x - 3) % P == 0

def g1_neg(pt):
    if pt is None:
        return None
    x, y = pt
    return (x, (-y) % P)

def g1_add(a, b):
    if a is None:
        return b
    if b is None:
        return a
    ax, ay = a
    bx, by = b
    if ax == bx:
        if (ay + by) % P == 0:
            return None
        # doubling
        lam = 3 * ax * ax * pow(2 * ay, -1, P) % P
    else:
        lam = (by - ay) * pow(bx - ax, -1, P) % P
    x3 = (lam * lam - ax - bx) % P
    y3 = (lam * (ax - x3) - ay) % P
    return (x3, y3)

def g1_mul(pt, k: int):
    k %= R
    acc = None
    add = pt
    while k:
        if k & 1:
            acc = g1_add(acc, add)
        add = g1_add(add, add)
        k >>= 1
    return acc

def msm(scalars, points):
    """Sum_i scalars[i] * points[i]  — the value best_multiexp computes."""
    acc = None
    for s, pt in zip(scalars, points):
        acc = g1_add(acc, g1_mul(pt, s))
    return acc


# ---------------------------------------------------------------- NTT (Fr)
def ntt(a, omega, log_n):
    """out[j] = Sum_i a[i] omega^(i j) mod r — what halo2's best_fft computes
    (its internal order: bit-reverse permutation + iterative DIT butterflies;
    the result is the plain DFT with the given omega)."""
    n = 1 << log_n
    assert len(a) == n
    out = list(a)
    # iterative Cooley-Tukey, bit-reversed input ordering (mirrors best_fft)
    if log_n > 0:
        for i in range(n):
            j = int(format(i, f"0{log_n}b")[::-1], 2)
            if j > i:
                out[i], out[j] = out[j], out[i]
    m = 1
    for _ in range(log_n):
        wm = pow(omega, n // (2 * m), R)
        for k in range(0, n, 2 * m):
            w = 1
            for j in range(m):
                t = w * out[k + j + m] % R
                u = out[k + j]
                out[k + j] = (u + t) % R
                out[k + j + m] = (u - t) % R
                w = w * wm % R
        m *= 2
    return out

def intt(a, omega, log_n):
    """Inverse: best_fft with omega^{-1}, then scale by n^{-1} (halo2
    EvaluationDomain::ifft semantics)."""
    n = 1 << log_n
    out = ntt(a, pow(omega, -1, R), log_n)
    n_inv = pow(n, -1, R)
    return [x * n_inv % R for x in out]

def distribute_powers(a, g):
    """a[i] *= g^i — halo2 EvaluationDomain coset pre/post multiply."""
    out = []
    cur = 1
    for x in a:
        out.append(x * cur % R)
        cur = cur * g % R
    return out


# ---------------------------------------------------------------- G2 (Fq2)
# BN254 G2: y^2 = x^3 + b2 over Fq2 = Fq[u]/(u^2 + 1), b2 = 3/(9+u) (the
# D-type sextic twist; halo2curves bn256::G2). Elements (c0, c1) = c0 + c1*u.
# Affine memory image (halo2curves G2Affine): x.c0 || x.c1 || y.c0 || y.c1,
# each 32 B LE Montgomery; identity encoded as 128 zero bytes.

def fq2_mul(a, b):
    a0, a1 = a
    b0, b1 = b
    return ((a0 * b0 - a1 * b1) % P, (a0 * b1 + a1 * b0) % P)

def fq2_inv(a):
    a0, a1 = a
    d = pow(a0 * a0 + a1 * a1, -1, P)
    return (a0 * d % P, (-a1) % P * d % P)

B2 = fq2_mul((3, 0), fq2_inv((9, 1)))

# Published BN254 G2 generator (the one halo2curves / EIP-197 use).
G2_GEN = (
    (10857046999023057135944570762232829481370756359578518086990519993285655852781,
     11559732032986387107991004021392285783925812861821192530917403151452391805634),
    (8495653923123431417604973247489272438418190587263600148770280649306958101930,
     4082367875863433681332203403145435568316851327593401208105741076214120093531),
)

def g2_is_on_curve(pt) -> bool:
    if pt is None:
        return True
    x, y = pt
    rhs = fq2_mul(fq2_mul(x, x), x)
    rhs = ((rhs[0] + B2[0]) % P, (rhs[1] + B2[1]) % P)
    lhs = fq2_mul(y, y)
    return lhs == rhs

assert g2_is_on_curve(G2_GEN), "G2 generator constants wrong"

def g2_neg(pt):
    if pt is None:
        return None
    x, y = pt
    return (x, ((-y[0]) % P, (-y[1]) % P))

def g2_add(a, b):
    if a is None:
        return b
    if b is None:
        return a
    (ax, ay), (bx, by) = a, b
    if ax == bx:
        if ((ay[0] + by[0]) % P, (ay[1] + by[1]) % P) == (0, 0):
            return None
        num = fq2_mul((3, 0), fq2_mul(ax, ax))
        lam = fq2_mul(num, fq2_inv(((2 * ay[0]) % P, (2 * ay[1]) % P)))
    else:
        num = ((by[0] - ay[0]) % P, (by[1] - ay[1]) % P)
        den = ((bx[0] - ax[0]) % P, (bx[1] - ax[1]) % P)
        lam = fq2_mul(num, fq2_inv(den))
    x3 = fq2_mul(lam, lam)
    x3 = ((x3[0] - ax[0] - bx[0]) % P, (x3[1] - ax[1] - bx[1]) % P)
    t = ((ax[0] - x3[0]) % P, (ax[1] - x3[1]) % P)
    y3 = fq2_mul(lam, t)
    y3 = ((y3[0] - ay[0]) % P, (y3[1] - ay[1]) % P)
    return (x3, y3)

def g2_mul(pt, k: int):
    k %= R
    acc = None
    add = pt
    while k:
        if k & 1:
            acc = g2_add(acc, add)
        add = g2_add(add, add)
        k >>= 1
    return acc

def g2_msm(scalars, points):
    acc = None
    for s, pt in zip(scalars, points):
        acc = g2_add(acc, g2_mul(pt, s))
    return acc

def g2_to_bytes(pt) -> bytes:
    if pt is None:
        return b"\x00" * 128
    x, y = pt
    return (to_mont_bytes(x[0], P) + to_mont_bytes(x[1], P) +
            to_mont_bytes(y[0], P) + to_mont_bytes(y[1], P))
