// g1.hpp — BN254 G1 point arithmetic (Jacobian over Fq), host + device.
//
// PRODUCT CODE. Formulas: EFD dbl-2009-l (a=0), madd-2007-bl, add-2007-bl
// with explicit degenerate-case handling. Jacobian identity: Z == 0.
// Affine memory image = halo2curves G1Affine: x||y Montgomery Fq, 64 bytes,
// identity encoded as (0, 0) (x=y=0 is not on y^2 = x^3 + 3).
//
// The accumulate primitives (g1j_dbl_ip / g1j_madd_ip / g1j_add_ip) mutate
// their accumulator in place: on gfx950 a g1_jac is 24 VGPRs, and the
// copy-in/copy-out style (o, p, q all distinct) makes hipcc spill to scratch
// in the MSM kernels — the in-place forms keep the whole working set in
// registers.
#pragma once
#include "ff.hpp"

struct g1_affine {
    fp256 x, y;  // Montgomery Fq; identity iff x==0 && y==0
};
struct g1_jac {
    fp256 X, Y, Z;  // identity iff Z==0
};

FF_HD bool g1a_is_inf(const g1_affine& p) { return ff_is_zero(p.x) && ff_is_zero(p.y); }
FF_HD void g1j_set_inf(g1_jac& p) {
    ff_set_one<Fq>(p.X);
    ff_set_one<Fq>(p.Y);
    ff_set_zero(p.Z);
}
FF_HD bool g1j_is_inf(const g1_jac& p) { return ff_is_zero(p.Z); }
FF_HD void g1j_from_affine(g1_jac& o, const g1_affine& p) {
    if (g1a_is_inf(p)) { g1j_set_inf(o); return; }
    o.X = p.x;
    o.Y = p.y;
    ff_set_one<Fq>(o.Z);
}
FF_HD void g1a_neg(g1_affine& o, const g1_affine& p) {
    o.x = p.x;
    ff_neg<Fq>(o.y, p.y);
}

// p = 2p, dbl-2009-l (a = 0), in place
FF_HD void g1j_dbl_ip(g1_jac& p) {
    if (g1j_is_inf(p)) return;
    fp256 A, B, C, D, E, F;
    ff_sqr<Fq>(A, p.X);
    ff_sqr<Fq>(B, p.Y);
    ff_sqr<Fq>(C, B);
    ff_add<Fq>(D, p.X, B);
    ff_sqr<Fq>(D, D);
    ff_sub<Fq>(D, D, A);
    ff_sub<Fq>(D, D, C);
    ff_add<Fq>(D, D, D);
    ff_add<Fq>(E, A, A);
    ff_add<Fq>(E, E, A);
    ff_sqr<Fq>(F, E);
    ff_mul<Fq>(p.Z, p.Y, p.Z);
    ff_add<Fq>(p.Z, p.Z, p.Z);
    ff_sub<Fq>(p.X, F, D);
    ff_sub<Fq>(p.X, p.X, D);
    ff_sub<Fq>(D, D, p.X);
    ff_mul<Fq>(D, E, D);
    ff_add<Fq>(C, C, C);
    ff_add<Fq>(C, C, C);
    ff_add<Fq>(C, C, C);
    ff_sub<Fq>(p.Y, D, C);
}

// p += q (q affine), madd-2007-bl, in place
FF_HD void g1j_madd_ip(g1_jac& p, const g1_affine& q) {
    if (g1a_is_inf(q)) return;
    if (g1j_is_inf(p)) { g1j_from_affine(p, q); return; }
    fp256 Z1Z1, U2, S2, H, HH, I, J, rr, V;
    ff_sqr<Fq>(Z1Z1, p.Z);
    ff_mul<Fq>(U2, q.x, Z1Z1);
    ff_mul<Fq>(S2, q.y, p.Z);
    ff_mul<Fq>(S2, S2, Z1Z1);
    ff_sub<Fq>(H, U2, p.X);
    ff_sub<Fq>(rr, S2, p.Y);
    if (ff_is_zero(H)) {
        if (ff_is_zero(rr)) { g1j_dbl_ip(p); return; }
        g1j_set_inf(p);
        return;
    }
    ff_add<Fq>(rr, rr, rr);
    ff_sqr<Fq>(HH, H);
    ff_add<Fq>(I, HH, HH);
    ff_add<Fq>(I, I, I);
    ff_mul<Fq>(J, H, I);
    ff_mul<Fq>(V, p.X, I);
    // Z3 = (Z1+H)^2 - Z1Z1 - HH   (consumes p.Z, H, Z1Z1, HH)
    ff_add<Fq>(p.Z, p.Z, H);
    ff_sqr<Fq>(p.Z, p.Z);
    ff_sub<Fq>(p.Z, p.Z, Z1Z1);
    ff_sub<Fq>(p.Z, p.Z, HH);
    // X3 = rr^2 - J - 2V
    ff_sqr<Fq>(H, rr);  // reuse H
    ff_sub<Fq>(H, H, J);
    ff_sub<Fq>(H, H, V);
    ff_sub<Fq>(H, H, V);
    // Y3 = rr*(V - X3) - 2*Y1*J
    ff_sub<Fq>(V, V, H);
    ff_mul<Fq>(V, rr, V);
    ff_mul<Fq>(J, p.Y, J);
    ff_add<Fq>(J, J, J);
    ff_sub<Fq>(p.Y, V, J);
    p.X = H;
}

// p += q (both Jacobian), add-2007-bl, in place
FF_HD void g1j_add_ip(g1_jac& p, const g1_jac& q) {
    if (g1j_is_inf(q)) return;
    if (g1j_is_inf(p)) { p = q; return; }
    fp256 Z1Z1, Z2Z2, U1, U2, S1, S2, H, I, J, rr, V;
    ff_sqr<Fq>(Z1Z1, p.Z);
    ff_sqr<Fq>(Z2Z2, q.Z);
    ff_mul<Fq>(U1, p.X, Z2Z2);
    ff_mul<Fq>(U2, q.X, Z1Z1);
    ff_mul<Fq>(S1, p.Y, q.Z);
    ff_mul<Fq>(S1, S1, Z2Z2);
    ff_mul<Fq>(S2, q.Y, p.Z);
    ff_mul<Fq>(S2, S2, Z1Z1);
    ff_sub<Fq>(H, U2, U1);
    ff_sub<Fq>(rr, S2, S1);
    if (ff_is_zero(H)) {
        if (ff_is_zero(rr)) { g1j_dbl_ip(p); return; }
        g1j_set_inf(p);
        return;
    }
    ff_add<Fq>(rr, rr, rr);
    ff_add<Fq>(I, H, H);
    ff_sqr<Fq>(I, I);
    ff_mul<Fq>(J, H, I);
    ff_mul<Fq>(V, U1, I);
    // Z3 = ((Z1+Z2)^2 - Z1Z1 - Z2Z2) * H
    ff_add<Fq>(p.Z, p.Z, q.Z);
    ff_sqr<Fq>(p.Z, p.Z);
    ff_sub<Fq>(p.Z, p.Z, Z1Z1);
    ff_sub<Fq>(p.Z, p.Z, Z2Z2);
    ff_mul<Fq>(p.Z, p.Z, H);
    // X3 = rr^2 - J - 2V
    ff_sqr<Fq>(H, rr);
    ff_sub<Fq>(H, H, J);
    ff_sub<Fq>(H, H, V);
    ff_sub<Fq>(H, H, V);
    // Y3 = rr*(V - X3) - 2*S1*J
    ff_sub<Fq>(V, V, H);
    ff_mul<Fq>(V, rr, V);
    ff_mul<Fq>(J, S1, J);
    ff_add<Fq>(J, J, J);
    ff_sub<Fq>(p.Y, V, J);
    p.X = H;
}

// ---- XYZZ coordinates (kept as a MEASURED NEGATIVE for bucket accumulation)
// x = X/ZZ, y = Y/ZZZ with ZZ^3 == ZZZ^2. Mixed add (mADD-2008-s) is
// 8M + 2S and ~4 add/sub vs Jacobian madd-2007-bl's 7M + 4S and ~7, BUT
// using it as the k_bucket_acc accumulator (with per-run g1xyzz_to_jac
// conversion) measured SLOWER on MI355X: 152 VGPRs / 3 waves/SIMD vs 84 /
// 4-5 for the Jacobian form — bucket_acc 2.31 -> 3.09 ms at n=2^20. The
// occupancy loss beats the ~9% multiply saving. Retained (bit-exact,
// unused on the hot path) for lower-pressure contexts. Identity: ZZ == 0.
struct g1_xyzz {
    fp256 X, Y, ZZ, ZZZ;
};

FF_HD void g1x_set_inf(g1_xyzz& p) {
    ff_set_one<Fq>(p.X);
    ff_set_one<Fq>(p.Y);
    ff_set_zero(p.ZZ);
    ff_set_zero(p.ZZZ);
}
FF_HD bool g1x_is_inf(const g1_xyzz& p) { return ff_is_zero(p.ZZ); }
FF_HD void g1x_from_affine(g1_xyzz& o, const g1_affine& p) {
    if (g1a_is_inf(p)) { g1x_set_inf(o); return; }
    o.X = p.x;
    o.Y = p.y;
    ff_set_one<Fq>(o.ZZ);
    ff_set_one<Fq>(o.ZZZ);
}

// p = 2p, dbl-2008-s-1 (a = 0), in place
FF_HD void g1x_dbl_ip(g1_xyzz& p) {
    if (g1x_is_inf(p)) return;
    fp256 U, V, W, S, M, t;
    ff_add<Fq>(U, p.Y, p.Y);   // U = 2Y
    ff_sqr<Fq>(V, U);          // V = U^2
    ff_mul<Fq>(W, U, V);       // W = U*V
    ff_mul<Fq>(S, p.X, V);     // S = X*V
    ff_sqr<Fq>(M, p.X);
    ff_add<Fq>(t, M, M);
    ff_add<Fq>(M, t, M);       // M = 3X^2
    ff_sqr<Fq>(t, M);
    ff_sub<Fq>(t, t, S);
    ff_sub<Fq>(t, t, S);       // X' = M^2 - 2S
    ff_sub<Fq>(S, S, t);
    ff_mul<Fq>(S, M, S);       // M*(S - X')
    ff_mul<Fq>(p.Y, W, p.Y);   // W*Y
    ff_sub<Fq>(p.Y, S, p.Y);   // Y' = M*(S-X') - W*Y
    p.X = t;
    ff_mul<Fq>(p.ZZ, p.ZZ, V);
    ff_mul<Fq>(p.ZZZ, p.ZZZ, W);
}

// p += q (q affine), mADD-2008-s, in place
FF_HD void g1x_madd_ip(g1_xyzz& p, const g1_affine& q) {
    if (g1a_is_inf(q)) return;
    if (g1x_is_inf(p)) { g1x_from_affine(p, q); return; }
    fp256 U2, S2, P, R, PP, PPP, Q, t;
    ff_mul<Fq>(U2, q.x, p.ZZ);
    ff_mul<Fq>(S2, q.y, p.ZZZ);
    ff_sub<Fq>(P, U2, p.X);
    ff_sub<Fq>(R, S2, p.Y);
    if (ff_is_zero(P)) {
        if (ff_is_zero(R)) { g1x_dbl_ip(p); return; }
        g1x_set_inf(p);
        return;
    }
    ff_sqr<Fq>(PP, P);
    ff_mul<Fq>(PPP, P, PP);
    ff_mul<Fq>(Q, p.X, PP);
    ff_sqr<Fq>(t, R);
    ff_sub<Fq>(t, t, PPP);
    ff_sub<Fq>(t, t, Q);
    ff_sub<Fq>(t, t, Q);       // X3 = R^2 - PPP - 2Q
    ff_sub<Fq>(Q, Q, t);       // Q - X3
    ff_mul<Fq>(Q, R, Q);
    ff_mul<Fq>(p.Y, p.Y, PPP);
    ff_sub<Fq>(p.Y, Q, p.Y);   // Y3 = R*(Q - X3) - Y1*PPP
    p.X = t;
    ff_mul<Fq>(p.ZZ, p.ZZ, PP);
    ff_mul<Fq>(p.ZZZ, p.ZZZ, PPP);
}

// xyzz -> Jacobian: Z = ZZZ/ZZ would need an inversion; instead pick
// Z = ZZ*ZZZ, so X_j = x*Z^2 = (X/ZZ)*(ZZ*ZZZ)^2 = X*ZZ*ZZZ^2 and
// Y_j = y*Z^3 = (Y/ZZZ)*(ZZ*ZZZ)^3 = Y*ZZ^3*ZZZ^2. Inversion-free,
// amortized once per run.
FF_HD void g1xyzz_to_jac(g1_jac& o, const g1_xyzz& p) {
    if (g1x_is_inf(p)) { g1j_set_inf(o); return; }
    fp256 zz2, zzz2, t;
    ff_sqr<Fq>(zzz2, p.ZZZ);              // ZZZ^2
    ff_mul<Fq>(o.Z, p.ZZ, p.ZZZ);         // Z = ZZ*ZZZ
    ff_mul<Fq>(t, p.ZZ, zzz2);            // ZZ*ZZZ^2
    ff_mul<Fq>(o.X, p.X, t);              // X_j
    ff_sqr<Fq>(zz2, p.ZZ);                // ZZ^2
    ff_mul<Fq>(t, zz2, p.ZZ);             // ZZ^3
    ff_mul<Fq>(t, t, zzz2);               // ZZ^3*ZZZ^2
    ff_mul<Fq>(o.Y, p.Y, t);              // Y_j
}

// ---- copy-style wrappers (host/ffi convenience) ----
FF_HD void g1j_dbl(g1_jac& o, const g1_jac& p) {
    o = p;
    g1j_dbl_ip(o);
}
FF_HD void g1j_add_affine(g1_jac& o, const g1_jac& p, const g1_affine& q) {
    o = p;
    g1j_madd_ip(o, q);
}
FF_HD void g1j_add(g1_jac& o, const g1_jac& p, const g1_jac& q) {
    o = p;
    g1j_add_ip(o, q);
}

// host-side normalization (field inversion — used once per MSM result)
FF_HD void g1j_to_affine(g1_affine& o, const g1_jac& p) {
    if (g1j_is_inf(p)) {
        ff_set_zero(o.x);
        ff_set_zero(o.y);
        return;
    }
    fp256 zi, zi2, zi3;
    ff_inv<Fq>(zi, p.Z);
    ff_sqr<Fq>(zi2, zi);
    ff_mul<Fq>(zi3, zi2, zi);
    ff_mul<Fq>(o.x, p.X, zi2);
    ff_mul<Fq>(o.y, p.Y, zi3);
}
