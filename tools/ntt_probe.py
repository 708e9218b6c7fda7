#!/usr/bin/env python3
"""NTT timing probe (GPU box): device-resident forward/inverse/coset NTT at
the BASELINE sizes. Not part of the bench contract — a tuning tool."""
import os
import sys
import time

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, HERE)
sys.path.insert(0, os.path.join(HERE, "oracle"))
import pywrap as oracle  # noqa: E402
from spectre_amd import SpectreGpu  # noqa: E402

R = 21888242871839275222246405745257275088548364400416034343698204186575808495617


def omega_for(log_n):
    g7 = oracle.fr_from_canonical((7).to_bytes(32, "little"))
    w = oracle.fr_pow(g7, ((R - 1) >> 28).to_bytes(32, "little"))
    for _ in range(28 - log_n):
        w = oracle.fr_mul(w, w)
    return w


def main():
    gpu = SpectreGpu([0])
    g = oracle.fr_from_canonical((5).to_bytes(32, "little"))
    sizes = ([int(a) for a in sys.argv[1:]] if len(sys.argv) > 1
             else [20, 22, 23, 24])
    for log_n in sizes:
        n = 1 << log_n
        data = oracle.gen_fr_vector(min(n, 1 << 20), 7)
        data = data * (n // (1 << min(log_n, 20)))
        w = omega_for(log_n)
        wi = oracle.fr_inv(w)
        d = gpu.malloc(32 * n)
        gpu.upload(d, data)
        gpu.ntt_device(d, log_n, w)  # warm (plan build)
        gpu.ntt_device(d, log_n, wi, inverse=True)
        reps = 10
        for name, kw in [("fwd", dict()), ("inv", dict(inverse=True)),
                         ("coset", dict(coset_gen=g))]:
            om = wi if kw.get("inverse") else w
            t0 = time.time()
            for _ in range(reps):
                gpu.ntt_device(d, log_n, om, **kw)
            dt = (time.time() - t0) / reps * 1e3
            gb = 2 * n * 32 / 1e9  # algorithmic: one read+write per pass pair
            print(f"2^{log_n} {name}: {dt:7.3f} ms  "
                  f"({2 * gb / dt * 1e3:7.1f} GB/s algorithmic 2-pass)")
        gpu.free(d)
    gpu.close()


if __name__ == "__main__":
    main()
