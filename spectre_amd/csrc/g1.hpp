// g1.hpp — BN254 G1 point arithmetic (Jacobian over Fq), host + device.
//
// PRODUCT CODE. Formulas: EFD dbl-2009-l (a=0), madd-2007-bl, add-2007-bl
// with explicit degenerate-case handling. Jacobian identity: Z == 0.
// Affine memory image = halo2curves G1Affine: x||y Montgomery Fq, 64 bytes,
// identity encoded as (0, 0) (x=y=0 is not on y^2 = x^3 + 3).
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

// dbl-2009-l, a = 0
FF_HD void g1j_dbl(g1_jac& o, const g1_jac& p) {
    if (g1j_is_inf(p)) { o = p; return; }
    fp256 A, B, Csq, D, E, F, t;
    ff_sqr<Fq>(A, p.X);
    ff_sqr<Fq>(B, p.Y);
    ff_sqr<Fq>(Csq, B);
    ff_add<Fq>(D, p.X, B);
    ff_sqr<Fq>(D, D);
    ff_sub<Fq>(D, D, A);
    ff_sub<Fq>(D, D, Csq);
    ff_add<Fq>(D, D, D);
    ff_add<Fq>(E, A, A);
    ff_add<Fq>(E, E, A);
    ff_sqr<Fq>(F, E);
    ff_mul<Fq>(t, p.Y, p.Z);
    ff_add<Fq>(o.Z, t, t);
    ff_sub<Fq>(o.X, F, D);
    ff_sub<Fq>(o.X, o.X, D);
    ff_sub<Fq>(t, D, o.X);
    ff_mul<Fq>(t, E, t);
    ff_add<Fq>(Csq, Csq, Csq);
    ff_add<Fq>(Csq, Csq, Csq);
    ff_add<Fq>(Csq, Csq, Csq);
    ff_sub<Fq>(o.Y, t, Csq);
}

// mixed add: o = p + q (q affine), madd-2007-bl + degenerate handling
FF_HD void g1j_add_affine(g1_jac& o, const g1_jac& p, const g1_affine& q) {
    if (g1a_is_inf(q)) { o = p; return; }
    if (g1j_is_inf(p)) { g1j_from_affine(o, q); return; }
    fp256 Z1Z1, U2, S2, H, HH, I, J, rr, V, t, Ynew;
    ff_sqr<Fq>(Z1Z1, p.Z);
    ff_mul<Fq>(U2, q.x, Z1Z1);
    ff_mul<Fq>(S2, q.y, p.Z);
    ff_mul<Fq>(S2, S2, Z1Z1);
    ff_sub<Fq>(H, U2, p.X);
    ff_sub<Fq>(rr, S2, p.Y);
    if (ff_is_zero(H)) {
        if (ff_is_zero(rr)) { g1j_dbl(o, p); return; }
        g1j_set_inf(o);
        return;
    }
    ff_add<Fq>(rr, rr, rr);
    ff_sqr<Fq>(HH, H);
    ff_add<Fq>(I, HH, HH);
    ff_add<Fq>(I, I, I);
    ff_mul<Fq>(J, H, I);
    ff_mul<Fq>(V, p.X, I);
    ff_sqr<Fq>(o.X, rr);
    ff_sub<Fq>(o.X, o.X, J);
    ff_sub<Fq>(o.X, o.X, V);
    ff_sub<Fq>(o.X, o.X, V);
    ff_sub<Fq>(t, V, o.X);
    ff_mul<Fq>(t, rr, t);
    ff_mul<Fq>(J, p.Y, J);
    ff_add<Fq>(J, J, J);
    ff_sub<Fq>(Ynew, t, J);
    ff_add<Fq>(t, p.Z, H);
    ff_sqr<Fq>(t, t);
    ff_sub<Fq>(t, t, Z1Z1);
    ff_sub<Fq>(o.Z, t, HH);
    o.Y = Ynew;
}

// general add: o = p + q, add-2007-bl + degenerate handling
FF_HD void g1j_add(g1_jac& o, const g1_jac& p, const g1_jac& q) {
    if (g1j_is_inf(p)) { o = q; return; }
    if (g1j_is_inf(q)) { o = p; return; }
    fp256 Z1Z1, Z2Z2, U1, U2, S1, S2, H, I, J, rr, V, t, Ynew;
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
        if (ff_is_zero(rr)) { g1j_dbl(o, p); return; }
        g1j_set_inf(o);
        return;
    }
    ff_add<Fq>(rr, rr, rr);
    ff_add<Fq>(I, H, H);
    ff_sqr<Fq>(I, I);
    ff_mul<Fq>(J, H, I);
    ff_mul<Fq>(V, U1, I);
    ff_sqr<Fq>(o.X, rr);
    ff_sub<Fq>(o.X, o.X, J);
    ff_sub<Fq>(o.X, o.X, V);
    ff_sub<Fq>(o.X, o.X, V);
    ff_sub<Fq>(t, V, o.X);
    ff_mul<Fq>(t, rr, t);
    ff_mul<Fq>(J, S1, J);
    ff_add<Fq>(J, J, J);
    ff_sub<Fq>(Ynew, t, J);
    ff_add<Fq>(t, p.Z, q.Z);
    ff_sqr<Fq>(t, t);
    ff_sub<Fq>(t, t, Z1Z1);
    ff_sub<Fq>(t, t, Z2Z2);
    ff_mul<Fq>(o.Z, t, H);
    o.Y = Ynew;
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
