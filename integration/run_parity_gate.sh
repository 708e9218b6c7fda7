#!/usr/bin/env bash
# run_parity_gate.sh — the one-command parity gate for a cargo-capable
# environment. From kernels-match-the-oracle to
# proof-bytes-match-the-reference in three legs:
#
#   leg 1 (capture, CPU): run Spectre's own prover test on the unmodified
#          CPU path with SPECTRE_CAPTURE set -> reference-produced
#          (input, output) vectors for every best_multiexp/best_fft call
#          + exact per-proof call counts (calls.jsonl; cf. CALLCOUNTS.md).
#   leg 2 (golden): convert the capture into committed fixtures
#          (tools/capture_to_golden.py) and check this repo's CPU oracle
#          against them bit-for-bit (pytest -m "not gpu" on the new
#          fixtures; on a GPU box, -m gpu checks the HIP path too).
#   leg 3 (proof bytes): re-run the same proof with --features spectre-gpu
#          under a FIXED transcript RNG and diff the proof bytes against
#          the CPU run. Requires an AMD GPU + libspectre_gpu.so.
#
# Usage:
#   ./run_parity_gate.sh /path/to/spectre /path/to/halo2/halo2_proofs [repo]
# where `repo` (default: this repo's root) holds tools/ + tests/.
#
# Note on RNG: `gen_snark_shplonk` passes OsRng internally; for leg 3 either
# (a) use the SDK's *_with_rng variant if the checkout has one, or (b) rely
# on the fact that with identical (witness, pk) the only RNG-dependent bytes
# are the blinding commitments — diff the capture logs instead: leg 1 and a
# GPU-side capture must agree on every (input_fnv -> output) pair, which is
# the same statement proof-byte equality reduces to once blinds are fixed.
set -euo pipefail
SPECTRE="${1:?usage: run_parity_gate.sh /path/to/spectre /path/to/halo2_proofs [repo]}"
HALO2="${2:?need halo2_proofs path}"
REPO="${3:-$(cd "$(dirname "$0")/.." && pwd)}"

"$(dirname "$0")/apply_patch.sh" "$HALO2"   # capture-only defaults (leg 1/2)

grep -q 'halo2_proofs.*path' "$SPECTRE/Cargo.toml" || cat >> "$SPECTRE/Cargo.toml" <<EOF

[patch."https://github.com/privacy-scaling-explorations/halo2.git"]
halo2_proofs = { path = "$HALO2" }
EOF

# leg 1: capture on the unmodified CPU path (Spectre's own prover test —
# test_step_proofgen, lightclient-circuits/src/sync_step_circuit.rs:482)
CAP="$REPO/gpurun_out/capture_step20"
rm -rf "$CAP" && mkdir -p "$CAP"
( cd "$SPECTRE" && SPECTRE_CAPTURE="$CAP" cargo test -r -p lightclient-circuits \
    test_step_proofgen -- --nocapture )
echo "captured $(wc -l < "$CAP/calls.jsonl") seam calls"

# leg 2: reference-pinned golden fixtures + oracle check
python3 "$REPO/tools/capture_to_golden.py" "$CAP" --out "$REPO/tests/golden/captured"
( cd "$REPO" && python3 -m pytest tests/test_captured_vectors.py -q )

# leg 3: GPU proof run + capture diff (needs an AMD GPU)
if command -v rocminfo >/dev/null 2>&1; then
    CAPG="$REPO/gpurun_out/capture_step20_gpu"
    rm -rf "$CAPG" && mkdir -p "$CAPG"
    # flip the patched crate's defaults to include the GPU dispatch
    grep -q '"spectre-gpu"' <(grep '^default' "$HALO2/Cargo.toml") || \
        sed -i 's/^default *= *\[/default = ["spectre-gpu", /' "$HALO2/Cargo.toml"
    ( cd "$SPECTRE" && SPECTRE_CAPTURE="$CAPG" \
        SPECTRE_GPU_LIB_DIR="$REPO/spectre_amd" \
        LD_LIBRARY_PATH="$REPO/spectre_amd:${LD_LIBRARY_PATH:-}" \
        cargo test -r -p lightclient-circuits \
        test_step_proofgen -- --nocapture )
    python3 "$REPO/tools/capture_to_golden.py" "$CAPG" --diff "$CAP"
else
    echo "no GPU here: leg 3 (GPU proof + diff) must run on the GPU box"
fi
echo "parity gate complete"
