#!/usr/bin/env python3
"""capture_to_golden.py — turn a SPECTRE_CAPTURE dump (produced by the
patched halo2_proofs running the UNMODIFIED CPU prover; integration/) into
reference-pinned golden fixtures and call-count evidence.

  --out DIR    write fixtures: cases small enough to commit (n <= 2^12)
               embedded fully (hex); larger calls summarized (fnv of inputs
               + full output). Also writes counts_summary.json — the exact
               per-proof MSM/NTT call counts that CALLCOUNTS.md derives.
  --diff OTHER compare two captures (e.g. CPU leg vs GPU leg of
               integration/run_parity_gate.sh): every (kind, input-id) must
               map to the same output. This is the statement proof-byte
               equality reduces to once blinding is fixed.

Record formats (integration/spectre.rs): msm_*.bin = "SPMSM1\\0\\0" u64 n,
n*32 scalars, n*64 bases, 64 out; fft_*.bin = "SPFFT1\\0\\0" u64 log_n,
32 omega, n*32 in, n*32 out. calls.jsonl has one line per seam call.
"""
import argparse
import collections
import json
import os
import struct
import sys


def read_calls(cap):
    path = os.path.join(cap, "calls.jsonl")
    with open(path) as f:
        return [json.loads(line) for line in f if line.strip()]


def read_msm_bin(path):
    with open(path, "rb") as f:
        magic = f.read(8)
        assert magic == b"SPMSM1\x00\x00", magic
        (n,) = struct.unpack("<Q", f.read(8))
        scalars = f.read(32 * n)
        bases = f.read(64 * n)
        out = f.read(64)
    return n, scalars, bases, out


def read_fft_bin(path):
    with open(path, "rb") as f:
        magic = f.read(8)
        assert magic == b"SPFFT1\x00\x00", magic
        (log_n,) = struct.unpack("<Q", f.read(8))
        omega = f.read(32)
        n = 1 << log_n
        inp = f.read(32 * n)
        outp = f.read(32 * n)
    return log_n, omega, inp, outp


def cmd_out(cap, outdir):
    calls = read_calls(cap)
    os.makedirs(outdir, exist_ok=True)
    msm_cases, fft_cases = [], []
    counts = collections.Counter()
    for c in calls:
        if c["kind"] == "msm":
            counts[("msm", c["n"])] += 1
            case = {"seq": c["seq"], "n": c["n"], "out": c["out"],
                    "scalars_fnv": c["scalars_fnv"], "bases_fnv": c["bases_fnv"]}
            if c.get("file") and c["n"] <= (1 << 12):
                n, sc, bs, out = read_msm_bin(os.path.join(cap, c["file"]))
                assert out.hex() == c["out"]
                case.update(scalars=sc.hex(), bases=bs.hex())
            msm_cases.append(case)
        else:
            counts[("fft", c["log_n"])] += 1
            case = {"seq": c["seq"], "log_n": c["log_n"], "omega": c["omega"],
                    "in_fnv": c["in_fnv"], "out_fnv": c["out_fnv"]}
            if c.get("file") and c["log_n"] <= 12:
                log_n, om, inp, outp = read_fft_bin(os.path.join(cap, c["file"]))
                case.update(input=inp.hex(), output=outp.hex())
            fft_cases.append(case)
    with open(os.path.join(outdir, "captured_msm.json"), "w") as f:
        json.dump({"provenance": "reference CPU prover via spectre-capture",
                   "cases": msm_cases}, f)
    with open(os.path.join(outdir, "captured_fft.json"), "w") as f:
        json.dump({"provenance": "reference CPU prover via spectre-capture",
                   "cases": fft_cases}, f)
    summary = [{"kind": k, "size": s, "count": n}
               for (k, s), n in sorted(counts.items())]
    with open(os.path.join(outdir, "counts_summary.json"), "w") as f:
        json.dump(summary, f, indent=1)
    print(f"wrote {len(msm_cases)} msm + {len(fft_cases)} fft cases; counts:")
    for row in summary:
        print(f"  {row['kind']} size {row['size']}: x{row['count']}")


def cmd_diff(cap_a, cap_b):
    """Align two captures by (kind, input identity) and require identical
    outputs. Input identity: full input FNV hashes (both captures hash the
    same raw bytes regardless of which backend computed the output)."""
    def index(cap):
        idx = {}
        for c in read_calls(cap):
            if c["kind"] == "msm":
                key = ("msm", c["n"], c["scalars_fnv"], c["bases_fnv"])
                val = c["out"]
            else:
                key = ("fft", c["log_n"], c["omega"], c["in_fnv"])
                val = c["out_fnv"]
            idx.setdefault(key, set()).add(val)
        return idx

    ia, ib = index(cap_a), index(cap_b)
    mismatch = missing = 0
    for key, outs_a in ia.items():
        outs_b = ib.get(key)
        if outs_b is None:
            missing += 1  # different blinding/RNG -> different inputs; ok
            continue
        if outs_a != outs_b:
            mismatch += 1
            print(f"MISMATCH {key}: {outs_a} vs {outs_b}")
    common = sum(1 for k in ia if k in ib)
    print(f"common inputs: {common}; output mismatches: {mismatch}; "
          f"only-in-A inputs (RNG-dependent): {missing}")
    if mismatch:
        sys.exit(1)
    if common == 0:
        print("WARNING: no common inputs — RNG not fixed; diff is vacuous")
        sys.exit(2)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("capture")
    ap.add_argument("--out")
    ap.add_argument("--diff")
    args = ap.parse_args()
    if args.out:
        cmd_out(args.capture, args.out)
    if args.diff:
        cmd_diff(args.diff, args.capture)
    if not args.out and not args.diff:
        ap.error("need --out and/or --diff")


if __name__ == "__main__":
    main()
