//! spectre.rs — GPU dispatch + capture shim for the PSE `halo2_proofs` fork.
//!
//! Installed by `integration/apply_patch.sh` as `src/spectre.rs` of the
//! halo2_proofs crate, together with thin wrappers around `best_multiexp` /
//! `best_fft` (integration/wrappers.rs). Two independent cargo features:
//!
//!   * `spectre-capture` — the ORIGINAL CPU path runs; every call through
//!     the seam is logged to `$SPECTRE_CAPTURE/` as (input, output) records
//!     plus one JSONL line per call. This pins reference-produced golden
//!     vectors and exact per-proof call counts (CALLCOUNTS.md) without any
//!     GPU present. Convert with tools/capture_to_golden.py.
//!   * `spectre-gpu` — BN254 G1 MSMs and Fr FFTs are dispatched to
//!     libspectre_gpu.so (include/spectre_gpu.h). With a seeded transcript
//!     RNG, proof bytes must be bit-identical to the CPU run
//!     (integration/run_parity_gate.sh).
//!
//! Both features sit strictly BELOW the reference's own operator API
//! (`AppCircuit`, lightclient-circuits/src/util/circuit.rs:163-218 — no
//! change above the seam).
//!
//! std-only; no new crate dependencies.

#![allow(dead_code)]

use std::fs::{File, OpenOptions};
use std::io::Write;
use std::sync::atomic::{AtomicU64, Ordering};
use std::sync::{Mutex, OnceLock};

// ---------------------------------------------------------------- capture

pub struct CaptureState {
    pub dir: String,
    pub seq: AtomicU64,
    pub jsonl: Mutex<File>,
    /// full (input,output) dumps per (kind, log2-size) class; further calls
    /// log hash + output only. Override: SPECTRE_CAPTURE_FULL.
    pub max_full: u64,
    pub full_counts: Mutex<std::collections::HashMap<(u8, u32), u64>>,
}

pub fn capture() -> Option<&'static CaptureState> {
    static S: OnceLock<Option<CaptureState>> = OnceLock::new();
    S.get_or_init(|| {
        let dir = std::env::var("SPECTRE_CAPTURE").ok()?;
        std::fs::create_dir_all(&dir).ok()?;
        let jsonl = OpenOptions::new()
            .create(true)
            .append(true)
            .open(format!("{dir}/calls.jsonl"))
            .ok()?;
        let max_full = std::env::var("SPECTRE_CAPTURE_FULL")
            .ok()
            .and_then(|v| v.parse().ok())
            .unwrap_or(4);
        Some(CaptureState {
            dir,
            seq: AtomicU64::new(0),
            jsonl: Mutex::new(jsonl),
            max_full,
            full_counts: Mutex::new(Default::default()),
        })
    })
    .as_ref()
}

/// FNV-1a 64 — content id for logged inputs (dedup/reference, not crypto).
pub fn fnv1a(bytes: &[u8]) -> u64 {
    let mut h: u64 = 0xcbf29ce484222325;
    for &b in bytes {
        h ^= b as u64;
        h = h.wrapping_mul(0x100000001b3);
    }
    h
}

fn hex(bytes: &[u8]) -> String {
    bytes.iter().map(|b| format!("{b:02x}")).collect()
}

/// Record one MSM crossing the seam. `scalars`/`bases` are the raw slice
/// memory (32 B LE Montgomery Fr / 64 B affine G1); `out_affine` is the
/// result's 64-B affine memory image, produced by the ORIGINAL CPU path.
pub fn log_msm(scalars: &[u8], bases: &[u8], out_affine: &[u8; 64]) {
    let Some(st) = capture() else { return };
    let n = (scalars.len() / 32) as u64;
    let seq = st.seq.fetch_add(1, Ordering::SeqCst);
    let class = (0u8, 64 - (n.max(1) - 1).leading_zeros());
    let full = {
        let mut fc = st.full_counts.lock().unwrap();
        let c = fc.entry(class).or_insert(0);
        *c += 1;
        n <= (1 << 12) || *c <= st.max_full
    };
    let file = if full {
        let name = format!("msm_{seq:06}.bin");
        if let Ok(mut f) = File::create(format!("{}/{name}", st.dir)) {
            // magic "SPMSM1", u64 n, scalars, bases, out
            let _ = f.write_all(b"SPMSM1\0\0");
            let _ = f.write_all(&n.to_le_bytes());
            let _ = f.write_all(scalars);
            let _ = f.write_all(bases);
            let _ = f.write_all(out_affine);
        }
        Some(name)
    } else {
        None
    };
    let line = format!(
        "{{\"seq\":{seq},\"kind\":\"msm\",\"n\":{n},\"scalars_fnv\":\"{:016x}\",\"bases_fnv\":\"{:016x}\",\"out\":\"{}\",\"file\":{}}}\n",
        fnv1a(scalars),
        fnv1a(bases),
        hex(out_affine),
        file.map_or("null".into(), |f| format!("\"{f}\"")),
    );
    let _ = st.jsonl.lock().unwrap().write_all(line.as_bytes());
}

/// Record one best_fft call. `input`/`output` are the raw 32 B/elem slice
/// memory before/after the ORIGINAL CPU path ran; omega is its 32-B image.
pub fn log_fft(input: &[u8], output: &[u8], omega: &[u8; 32], log_n: u32) {
    let Some(st) = capture() else { return };
    let seq = st.seq.fetch_add(1, Ordering::SeqCst);
    let class = (1u8, log_n);
    let full = {
        let mut fc = st.full_counts.lock().unwrap();
        let c = fc.entry(class).or_insert(0);
        *c += 1;
        log_n <= 12 || *c <= st.max_full
    };
    let file = if full {
        let name = format!("fft_{seq:06}.bin");
        if let Ok(mut f) = File::create(format!("{}/{name}", st.dir)) {
            let _ = f.write_all(b"SPFFT1\0\0");
            let _ = f.write_all(&(log_n as u64).to_le_bytes());
            let _ = f.write_all(omega);
            let _ = f.write_all(input);
            let _ = f.write_all(output);
        }
        Some(name)
    } else {
        None
    };
    let line = format!(
        "{{\"seq\":{seq},\"kind\":\"fft\",\"log_n\":{log_n},\"omega\":\"{}\",\"in_fnv\":\"{:016x}\",\"out_fnv\":\"{:016x}\",\"file\":{}}}\n",
        hex(omega),
        fnv1a(input),
        fnv1a(output),
        file.map_or("null".into(), |f| format!("\"{f}\"")),
    );
    let _ = st.jsonl.lock().unwrap().write_all(line.as_bytes());
}

// ---------------------------------------------------------------- gpu ffi

#[cfg(feature = "spectre-gpu")]
pub mod gpu {
    use std::os::raw::{c_char, c_int};
    use std::sync::OnceLock;

    #[repr(C)]
    pub struct Ctx {
        _p: [u8; 0],
    }
    extern "C" {
        pub fn spectre_gpu_init(ndev: c_int, ids: *const c_int) -> *mut Ctx;
        pub fn spectre_gpu_last_error() -> *const c_char;
        pub fn spectre_gpu_msm_g1(
            ctx: *mut Ctx,
            bases_id: u64,
            bases: *const u8,
            scalars: *const u8,
            n: u64,
            flags: u32,
            num_gpus: c_int,
            out_affine: *mut u8,
        ) -> c_int;
        pub fn spectre_gpu_ntt_fr(
            ctx: *mut Ctx,
            data: *mut u8,
            log_n: u32,
            omega: *const u8,
            inverse: c_int,
            coset_gen: *const u8,
        ) -> c_int;
    }

    pub fn ctx() -> *mut Ctx {
        static CTX: OnceLock<usize> = OnceLock::new();
        *CTX.get_or_init(|| {
            let c = unsafe { spectre_gpu_init(0, std::ptr::null()) };
            assert!(!c.is_null(), "spectre_gpu_init failed (GPU required; no CPU fallback)");
            c as usize
        }) as *mut Ctx
    }

    /// Stable per-SRS cache key (upload-once across a proof's ~45 commits):
    /// the base slice's address+len is stable for a loaded ParamsKZG.
    pub fn bases_id(b: &[u8]) -> u64 {
        (b.as_ptr() as u64) ^ ((b.len() as u64) << 1) | 1
    }

    pub fn last_error() -> String {
        unsafe {
            std::ffi::CStr::from_ptr(spectre_gpu_last_error())
                .to_string_lossy()
                .into_owned()
        }
    }
}
