// wrappers.rs — appended to the halo2_proofs source file that defines
// `best_multiexp` / `best_fft` AFTER apply_patch.sh renamed the originals
// to `best_multiexp_cpu` / `best_fft_cpu`. The wrappers keep the exact
// public signatures (the seam contract); everything above
// the seam (AppCircuit, create_proof) compiles unchanged.
//
// Dispatch rules:
//   * spectre-capture: run the ORIGINAL CPU path, then log the
//     (input, output) pair + one calls.jsonl line (crate::spectre).
//   * spectre-gpu: BN254 G1 / Fr work of size >= 2^10 goes to
//     libspectre_gpu.so; anything else falls through to the CPU original.
//   * both: capture logs the GPU result (useful for on-box diffing), but
//     the parity gate runs them separately (run_parity_gate.sh).

pub fn best_multiexp<C: CurveAffine>(coeffs: &[C::Scalar], bases: &[C]) -> C::Curve {
    let n = coeffs.len().min(bases.len());
    let is_bn254_g1 = core::any::TypeId::of::<C>()
        == core::any::TypeId::of::<halo2curves::bn256::G1Affine>();

    #[cfg(feature = "spectre-gpu")]
    if is_bn254_g1 && n >= (1 << 10) {
        let sc = unsafe { core::slice::from_raw_parts(coeffs.as_ptr() as *const u8, n * 32) };
        let bs = unsafe { core::slice::from_raw_parts(bases.as_ptr() as *const u8, n * 64) };
        let mut out = [0u8; 64];
        let rc = unsafe {
            crate::spectre::gpu::spectre_gpu_msm_g1(
                crate::spectre::gpu::ctx(),
                crate::spectre::gpu::bases_id(bs),
                bs.as_ptr(),
                sc.as_ptr(),
                n as u64,
                /*SPECTRE_SCALARS_MONTGOMERY*/ 0,
                1,
                out.as_mut_ptr(),
            )
        };
        assert_eq!(rc, 0, "spectre_gpu_msm_g1: {}", crate::spectre::gpu::last_error());
        // 64-B affine memory image (identity = zeros) -> C::Curve
        let pt: halo2curves::bn256::G1Affine = unsafe { core::mem::transmute_copy(&out) };
        let res: halo2curves::bn256::G1 = pt.into();
        #[cfg(feature = "spectre-capture")]
        crate::spectre::log_msm(
            unsafe { core::slice::from_raw_parts(coeffs.as_ptr() as *const u8, n * 32) },
            unsafe { core::slice::from_raw_parts(bases.as_ptr() as *const u8, n * 64) },
            &out,
        );
        return unsafe { core::mem::transmute_copy(&res) };
    }

    let result = best_multiexp_cpu(coeffs, bases);

    #[cfg(feature = "spectre-capture")]
    if is_bn254_g1 && crate::spectre::capture().is_some() {
        use halo2curves::group::Curve;
        let aff = result.to_affine();
        let out: &[u8; 64] = unsafe { &*(&aff as *const _ as *const [u8; 64]) };
        crate::spectre::log_msm(
            unsafe { core::slice::from_raw_parts(coeffs.as_ptr() as *const u8, n * 32) },
            unsafe { core::slice::from_raw_parts(bases.as_ptr() as *const u8, n * 64) },
            out,
        );
    }
    result
}

// ---SPLIT-FFT--- (apply_patch.sh splits here when best_fft lives in a
// different source file from best_multiexp)
pub fn best_fft<Scalar: Field, G: FftGroup<Scalar>>(a: &mut [G], omega: Scalar, log_n: u32) {
    let is_fr32 = core::mem::size_of::<G>() == 32 && core::mem::size_of::<Scalar>() == 32
        && core::any::TypeId::of::<Scalar>()
            == core::any::TypeId::of::<halo2curves::bn256::Fr>();

    #[cfg(feature = "spectre-capture")]
    let cap_input: Option<Vec<u8>> = if is_fr32 && crate::spectre::capture().is_some() {
        Some(unsafe {
            core::slice::from_raw_parts(a.as_ptr() as *const u8, a.len() * 32).to_vec()
        })
    } else {
        None
    };

    #[cfg(feature = "spectre-gpu")]
    if is_fr32 && log_n >= 10 {
        let data =
            unsafe { core::slice::from_raw_parts_mut(a.as_mut_ptr() as *mut u8, a.len() * 32) };
        let om: &[u8; 32] = unsafe { &*(&omega as *const _ as *const [u8; 32]) };
        let rc = unsafe {
            crate::spectre::gpu::spectre_gpu_ntt_fr(
                crate::spectre::gpu::ctx(),
                data.as_mut_ptr(),
                log_n,
                om.as_ptr(),
                0,
                core::ptr::null(),
            )
        };
        assert_eq!(rc, 0, "spectre_gpu_ntt_fr: {}", crate::spectre::gpu::last_error());
        #[cfg(feature = "spectre-capture")]
        if let Some(inp) = cap_input {
            let outp =
                unsafe { core::slice::from_raw_parts(a.as_ptr() as *const u8, a.len() * 32) };
            let om2: &[u8; 32] = unsafe { &*(&omega as *const _ as *const [u8; 32]) };
            crate::spectre::log_fft(&inp, outp, om2, log_n);
        }
        return;
    }

    best_fft_cpu(a, omega, log_n);

    #[cfg(feature = "spectre-capture")]
    if let Some(inp) = cap_input {
        let outp = unsafe { core::slice::from_raw_parts(a.as_ptr() as *const u8, a.len() * 32) };
        let om: &[u8; 32] = unsafe { &*(&omega as *const _ as *const [u8; 32]) };
        crate::spectre::log_fft(&inp, outp, om, log_n);
    }
}
