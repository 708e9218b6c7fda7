#!/usr/bin/env bash
# apply_patch.sh — install the spectre GPU/capture shim into a checkout of
# the PSE halo2_proofs fork (the crate halo2-base's `halo2-pse` feature
# pulls). Rename-and-wrap instead of a line-anchored diff so the patch
# survives upstream drift: the only assumptions are the two public seam
# signatures (`pub fn best_multiexp<`, `pub fn best_fft<`), which ARE the
# drop-in contract.
#
# Usage:  ./apply_patch.sh /path/to/halo2/halo2_proofs
# After:  build Spectre with
#   [patch."https://github.com/privacy-scaling-explorations/halo2.git"]
#   halo2_proofs = { path = ".../halo2_proofs" }
# and `--features halo2_proofs/spectre-capture` (capture, CPU-only) or
# `--features halo2_proofs/spectre-gpu` (GPU dispatch; set
# SPECTRE_GPU_LIB_DIR to the directory holding libspectre_gpu.so).
set -euo pipefail
CRATE="${1:?usage: apply_patch.sh /path/to/halo2_proofs}"
HERE="$(cd "$(dirname "$0")" && pwd)"
SRC="$CRATE/src"
[ -f "$CRATE/Cargo.toml" ] || { echo "no Cargo.toml under $CRATE"; exit 1; }

grep -q "pub mod spectre" "$SRC/lib.rs" && { echo "already applied"; exit 0; }

msm_file=$(grep -rl "pub fn best_multiexp<" "$SRC" | head -1)
fft_file=$(grep -rl "pub fn best_fft<" "$SRC" | head -1)
[ -n "$msm_file" ] || { echo "best_multiexp definition not found"; exit 1; }
[ -n "$fft_file" ] || { echo "best_fft definition not found"; exit 1; }
echo "best_multiexp in $msm_file; best_fft in $fft_file"

# 1. rename the originals (bodies untouched — they stay the CPU path)
sed -i 's/pub fn best_multiexp</pub fn best_multiexp_cpu</' "$msm_file"
sed -i 's/pub fn best_fft</pub fn best_fft_cpu</' "$fft_file"

# 2. append the wrappers (exact original signatures)
split_line=$(grep -n -- "---SPLIT-FFT---" "$HERE/wrappers.rs" | cut -d: -f1)
if [ "$msm_file" = "$fft_file" ]; then
    grep -v -- "---SPLIT-FFT---" "$HERE/wrappers.rs" >> "$msm_file"
else
    head -n $((split_line - 1)) "$HERE/wrappers.rs" >> "$msm_file"
    tail -n +$((split_line + 1)) "$HERE/wrappers.rs" >> "$fft_file"
fi

# 3. the shim module
cp "$HERE/spectre.rs" "$SRC/spectre.rs"
printf '\npub mod spectre;\n' >> "$SRC/lib.rs"

# 4. cargo features (+ link script for the gpu feature).
# halo2_proofs is a TRANSITIVE dependency of Spectre (via halo2-base), and
# cargo cannot enable a transitive dep's features from the CLI — so the
# shim features are toggled as DEFAULTS of the patched (local) crate:
#   apply_patch.sh CRATE            -> default = capture only (CPU-safe;
#                                      capture is inert without
#                                      SPECTRE_CAPTURE set at run time)
#   apply_patch.sh CRATE gpu        -> default = capture + gpu dispatch
MODE="${2:-capture}"
ADD='"spectre-capture"'
[ "$MODE" = gpu ] && ADD='"spectre-capture", "spectre-gpu"'
if grep -q '^\[features\]' "$CRATE/Cargo.toml"; then
    sed -i "/^\[features\]/a spectre-gpu = []\nspectre-capture = []" "$CRATE/Cargo.toml"
else
    printf '\n[features]\nspectre-gpu = []\nspectre-capture = []\n' >> "$CRATE/Cargo.toml"
fi
# MERGE into any existing default list (upstream defaults must survive);
# TOML tolerates the trailing comma when the list was empty.
if grep -q '^default *= *\[' "$CRATE/Cargo.toml"; then
    sed -i "s/^default *= *\[/default = [$ADD, /" "$CRATE/Cargo.toml"
else
    sed -i "/^\[features\]/a default = [$ADD]" "$CRATE/Cargo.toml"
fi
if [ ! -f "$CRATE/build.rs" ]; then
    cat > "$CRATE/build.rs" <<'EOF'
fn main() {
    // link libspectre_gpu.so only when the gpu feature is on
    if std::env::var("CARGO_FEATURE_SPECTRE_GPU").is_ok() {
        let dir = std::env::var("SPECTRE_GPU_LIB_DIR")
            .expect("set SPECTRE_GPU_LIB_DIR to the dir containing libspectre_gpu.so");
        println!("cargo:rustc-link-search=native={dir}");
        println!("cargo:rustc-link-lib=dylib=spectre_gpu");
    }
}
EOF
fi
echo "applied (defaults include spectre-capture$([ "$MODE" = gpu ] && echo /spectre-gpu)); SPECTRE_CAPTURE env gates logging at run time."
