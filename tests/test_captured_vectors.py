"""Tests for the capture -> golden pipeline (integration/ +
tools/capture_to_golden.py).

Two layers:
  * selftest (always runs): synthesize a capture dir in the exact binary/
    JSONL format integration/spectre.rs writes (using the CPU oracle as the
    stand-in prover), run the converter, and check the fixtures + the diff
    logic. This keeps the one-command parity harness exercised even though
    no cargo exists in this image.
  * captured fixtures (runs when tests/golden/captured/ exists — i.e. after
    integration/run_parity_gate.sh leg 1 ran in a cargo-capable env):
    checks the CPU oracle (and, under -m gpu, the HIP path) bit-for-bit
    against reference-produced vectors.
"""
import json
import os
import struct
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CAPTURED = os.path.join(REPO, "tests", "golden", "captured")
TOOL = os.path.join(REPO, "tools", "capture_to_golden.py")


def fnv1a(b: bytes) -> str:
    h = 0xCBF29CE484222325
    for x in b:
        h = ((h ^ x) * 0x100000001B3) & 0xFFFFFFFFFFFFFFFF
    return f"{h:016x}"


def write_capture(tmp, oracle, seed):
    """Synthesize a capture dir: 2 small MSMs + 1 small FFT, formats per
    integration/spectre.rs."""
    calls = []
    seq = 0
    for n in (64, 257):
        sc_can, bs = oracle.gen_msm_inputs(n, seed + n, fast=True)
        # the Rust shim logs Montgomery scalar images; build them
        sc = b"".join(oracle.fr_from_canonical(sc_can[32 * i:32 * (i + 1)])
                      for i in range(n))
        out = oracle.msm(bs, sc, n, scalars_canonical=False)
        name = f"msm_{seq:06}.bin"
        with open(os.path.join(tmp, name), "wb") as f:
            f.write(b"SPMSM1\x00\x00" + struct.pack("<Q", n) + sc + bs + out)
        calls.append({"seq": seq, "kind": "msm", "n": n,
                      "scalars_fnv": fnv1a(sc), "bases_fnv": fnv1a(bs),
                      "out": out.hex(), "file": name})
        seq += 1
    log_n = 8
    fix = json.load(open(os.path.join(REPO, "tests", "golden", "ntt.json")))
    omega12 = bytes.fromhex(
        [c for c in fix["seeded_cases"] if c["log_n"] == 12][0]["omega_mont"])
    omega = omega12
    for _ in range(12 - log_n):
        omega = oracle.fr_mul(omega, omega)
    inp = oracle.gen_fr_vector(1 << log_n, seed)
    outp = oracle.ntt(inp, log_n, omega)
    name = f"fft_{seq:06}.bin"
    with open(os.path.join(tmp, name), "wb") as f:
        f.write(b"SPFFT1\x00\x00" + struct.pack("<Q", log_n) + omega + inp + outp)
    calls.append({"seq": seq, "kind": "fft", "log_n": log_n,
                  "omega": omega.hex(), "in_fnv": fnv1a(inp),
                  "out_fnv": fnv1a(outp), "file": name})
    with open(os.path.join(tmp, "calls.jsonl"), "w") as f:
        for c in calls:
            f.write(json.dumps(c) + "\n")
    return calls


def test_capture_pipeline_selftest(tmp_path, oracle):
    cap = tmp_path / "cap"
    cap.mkdir()
    write_capture(str(cap), oracle, 1234)
    out = tmp_path / "golden"
    subprocess.run([sys.executable, TOOL, str(cap), "--out", str(out)],
                   check=True, capture_output=True)
    msm = json.load(open(out / "captured_msm.json"))
    fft = json.load(open(out / "captured_fft.json"))
    counts = json.load(open(out / "counts_summary.json"))
    assert len(msm["cases"]) == 2 and len(fft["cases"]) == 1
    assert {(r["kind"], r["size"]): r["count"] for r in counts} == {
        ("msm", 64): 1, ("msm", 257): 1, ("fft", 8): 1}
    # every embedded case replays bit-exactly on the oracle
    for c in msm["cases"]:
        got = oracle.msm(bytes.fromhex(c["bases"]),
                         bytes.fromhex(c["scalars"]), c["n"],
                         scalars_canonical=False)
        assert got.hex() == c["out"]
    for c in fft["cases"]:
        got = oracle.ntt(bytes.fromhex(c["input"]), c["log_n"],
                         bytes.fromhex(c["omega"]))
        assert got.hex() == c["output"]
    # diff: identical capture agrees with itself ...
    r = subprocess.run([sys.executable, TOOL, str(cap), "--diff", str(cap)],
                       capture_output=True, text=True)
    assert r.returncode == 0, r.stdout + r.stderr
    # ... and a corrupted output is caught
    cap2 = tmp_path / "cap2"
    cap2.mkdir()
    lines = open(cap / "calls.jsonl").read().splitlines()
    bad = json.loads(lines[0])
    bad["out"] = "00" * 64
    with open(cap2 / "calls.jsonl", "w") as f:
        f.write(json.dumps(bad) + "\n" + "\n".join(lines[1:]) + "\n")
    r = subprocess.run([sys.executable, TOOL, str(cap2), "--diff", str(cap)],
                       capture_output=True, text=True)
    assert r.returncode == 1 and "MISMATCH" in r.stdout


# ---- reference-captured fixtures (present only after leg 1 of
# integration/run_parity_gate.sh ran in a cargo environment) ----
needs_capture = pytest.mark.skipif(
    not os.path.exists(os.path.join(CAPTURED, "captured_msm.json")),
    reason="no reference capture committed (needs a cargo env; see "
           "integration/run_parity_gate.sh)")


@needs_capture
def test_oracle_vs_captured_msm(oracle):
    msm = json.load(open(os.path.join(CAPTURED, "captured_msm.json")))
    done = 0
    for c in msm["cases"]:
        if "scalars" not in c:
            continue
        got = oracle.msm(bytes.fromhex(c["bases"]),
                         bytes.fromhex(c["scalars"]), c["n"],
                         scalars_canonical=False)
        assert got.hex() == c["out"], f"seq {c['seq']}"
        done += 1
    assert done > 0


@needs_capture
def test_oracle_vs_captured_fft(oracle):
    fft = json.load(open(os.path.join(CAPTURED, "captured_fft.json")))
    done = 0
    for c in fft["cases"]:
        if "input" not in c:
            continue
        got = oracle.ntt(bytes.fromhex(c["input"]), c["log_n"],
                         bytes.fromhex(c["omega"]))
        assert got.hex() == c["output"], f"seq {c['seq']}"
        done += 1
    assert done > 0


@needs_capture
@pytest.mark.gpu
def test_gpu_vs_captured(oracle, gpu):
    msm = json.load(open(os.path.join(CAPTURED, "captured_msm.json")))
    for c in msm["cases"]:
        if "scalars" not in c:
            continue
        got = gpu.msm(bytes.fromhex(c["bases"]), bytes.fromhex(c["scalars"]),
                      c["n"], canonical=False)
        assert got.hex() == c["out"], f"seq {c['seq']}"
    fft = json.load(open(os.path.join(CAPTURED, "captured_fft.json")))
    for c in fft["cases"]:
        if "input" not in c:
            continue
        got = gpu.ntt(bytes.fromhex(c["input"]), c["log_n"],
                      bytes.fromhex(c["omega"]))
        assert got.hex() == c["output"], f"seq {c['seq']}"
