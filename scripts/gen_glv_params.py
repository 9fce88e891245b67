#!/usr/bin/env python3
"""Generate include/bn254_glv.h — GLV endomorphism constants for BN254 G1.

BN254 has j-invariant 0 (y^2 = x^3 + 3), so phi(x, y) = (beta*x, y) with
beta a primitive cube root of unity in Fq is an endomorphism acting as
scalar multiplication by lambda (a cube root of unity in Fr):
phi(P) = lambda * P for P in G1.  Scalars decompose as
k = k1 + lambda*k2 (mod r) with |k1|, |k2| < 2^128 via the rounding method
over a short lattice basis.  Everything below is DERIVED from the curve
moduli and re-verified numerically before emission.
"""
import sys

q = 21888242871839275222246405745257275088696311157297823662689037894645226208583
r = 21888242871839275222246405745257275088548364400416034343698204186575808495617


def tonelli(n, p):
    assert pow(n, (p - 1) // 2, p) == 1
    if p % 4 == 3:
        return pow(n, (p + 1) // 4, p)
    s, e = p - 1, 0
    while s % 2 == 0:
        s //= 2
        e += 1
    for z in range(2, 1000):
        if pow(z, (p - 1) // 2, p) == p - 1:
            break
    x = pow(n, (s + 1) // 2, p)
    b = pow(n, s, p)
    g = pow(z, s, p)
    rr = e
    while True:
        t = b
        m = 0
        for m in range(rr):
            if t == 1:
                break
            t = pow(t, 2, p)
        if m == 0:
            return x
        gs = pow(g, 2 ** (rr - m - 1), p)
        g = gs * gs % p
        x = x * gs % p
        b = b * g % p
        rr = m


def cube_root_of_unity(p):
    # roots of x^2 + x + 1: (-1 +- sqrt(-3)) / 2
    s = tonelli(p - 3, p)
    inv2 = pow(2, -1, p)
    w1 = (-1 + s) * inv2 % p
    assert pow(w1, 3, p) == 1 and w1 != 1
    return w1


beta = cube_root_of_unity(q)
lam = cube_root_of_unity(r)

# pick the (beta, lambda) pairing such that phi(P) == lambda*P on the curve
def ec_add(P, Q):
    if P is None:
        return Q
    if Q is None:
        return P
    x1, y1 = P
    x2, y2 = Q
    if x1 == x2 and (y1 + y2) % q == 0:
        return None
    if P == Q:
        m = 3 * x1 * x1 * pow(2 * y1, -1, q) % q
    else:
        m = (y2 - y1) * pow(x2 - x1, -1, q) % q
    x3 = (m * m - x1 - x2) % q
    return (x3, (m * (x1 - x3) - y1) % q)


def ec_mul(k, P):
    R = None
    while k:
        if k & 1:
            R = ec_add(R, P)
        P = ec_add(P, P)
        k >>= 1
    return R


G = (1, 2)  # BN254 G1 generator
cands = [(beta, lam), (beta, lam * lam % r), (beta * beta % q, lam),
         (beta * beta % q, lam * lam % r)]
beta, lam = next((b, l) for b, l in cands
                 if ec_mul(l, G) == (b * G[0] % q, G[1]))
assert ec_mul(lam, G) == (beta % q * G[0] % q, G[1])

# short lattice basis for (r, lambda): rows (a, b) with a + b*lambda = 0 mod r
# via truncated extended Euclid (GLV01)
def short_basis(n, l):
    s, t, ss, tt = n, 0, l, 1
    seq = []
    while ss != 0:
        qq = s // ss
        s, ss = ss, s - qq * ss
        t, tt = tt, t - qq * tt
        seq.append((s, t))
    import math
    half = math.isqrt(n) + 1
    for i in range(len(seq) - 1):
        if seq[i][0] >= half > seq[i + 1][0]:
            a1, b1 = seq[i + 1][0], -seq[i + 1][1]
            a2, b2 = seq[i][0], -seq[i][1]
            a3, b3 = (seq[i + 2][0], -seq[i + 2][1]) if i + 2 < len(seq) else (None, None)
            if a3 is not None and a3 * a3 + b3 * b3 < a2 * a2 + b2 * b2:
                a2, b2 = a3, b3
            return (a1, b1), (a2, b2)
    raise RuntimeError


(a1, b1), (a2, b2) = short_basis(r, lam)
# normalize row signs so both fixed-point multipliers are non-negative
if b1 > 0:
    a1, b1 = -a1, -b1
if b2 < 0:
    a2, b2 = -a2, -b2
for a, b in [(a1, b1), (a2, b2)]:
    assert (a + b * lam) % r == 0
    assert abs(a) < 2**128 and abs(b) < 2**128

# rounding method: c1 = round(b2*k/r), c2 = round(-b1*k/r);
# k1 = k - c1*a1 - c2*a2, k2 = -c1*b1 - c2*b2; k1 + k2*lam == k mod r
# fixed-point: g1 = round(2^384 * b2 / r), g2 = round(2^384 * -b1 / r)
SH = 320
g1 = (b2 * (1 << SH) + r // 2) // r
g2 = (-b1 * (1 << SH) + r // 2) // r
assert 0 <= g1 < 2**256 and 0 <= g2 < 2**256


def decompose(k):
    c1 = (g1 * k + (1 << (SH - 1))) >> SH
    c2 = (g2 * k + (1 << (SH - 1))) >> SH
    k1 = k - c1 * a1 - c2 * a2
    k2 = -c1 * b1 - c2 * b2
    return k1, k2


import random
rng = random.Random(7)
maxk1 = maxk2 = 0
for _ in range(20000):
    k = rng.randrange(r)
    k1, k2 = decompose(k)
    assert (k1 + k2 * lam) % r == k % r
    maxk1 = max(maxk1, abs(k1))
    maxk2 = max(maxk2, abs(k2))
for k in [0, 1, 2, r - 1, r // 2, lam, r - lam]:
    k1, k2 = decompose(k)
    assert (k1 + k2 * lam) % r == k % r
    maxk1 = max(maxk1, abs(k1))
    maxk2 = max(maxk2, abs(k2))
assert maxk1 < 2**128 and maxk2 < 2**128, (maxk1.bit_length(), maxk2.bit_length())
# spot-check the full identity on the curve
for _ in range(5):
    k = rng.randrange(r)
    k1, k2 = decompose(k)
    P = ec_mul(rng.randrange(1, 1 << 60), G)
    phiP = (beta * P[0] % q, P[1])
    lhs = ec_mul(k, P)
    t1 = ec_mul(abs(k1), P)
    if k1 < 0:
        t1 = (t1[0], q - t1[1])
    t2 = ec_mul(abs(k2), phiP)
    if k2 < 0:
        t2 = (t2[0], q - t2[1])
    assert ec_add(t1, t2) == lhs

Rq = 1 << 256


def limbs(x, n=4):
    return [(x >> (64 * i)) & ((1 << 64) - 1) for i in range(n)]


def fmt(name, x, n=4, comment=""):
    ls = limbs(x, n)
    body = ", ".join(f"0x{l:016x}ull" for l in ls)
    c = f"  /* {comment} */" if comment else ""
    return f"#define {name} {{{body}}}{c}"


def signed_parts(v):
    return (1 if v < 0 else 0), abs(v)


s_a1, m_a1 = signed_parts(a1)
s_b1, m_b1 = signed_parts(b1)
s_a2, m_a2 = signed_parts(a2)
s_b2, m_b2 = signed_parts(b2)

lines = [
    "/* BN254 G1 GLV endomorphism constants — GENERATED by",
    " * scripts/gen_glv_params.py (derived from the curve moduli and",
    " * numerically re-verified there: phi(P) = lambda*P, decomposition",
    " * round-trips, |k1|,|k2| < 2^128 over 20k random scalars + edges).",
    " */",
    "#ifndef BN254_GLV_H",
    "#define BN254_GLV_H",
    "",
    fmt("GLV_BETA_MONT", beta * Rq % q, 4, "cube root of unity in Fq (Montgomery)"),
    fmt("GLV_LAMBDA_PLAIN", lam, 4, "cube root of unity in Fr (plain)"),
    f"/* short basis: (a1,b1)=({a1},{b1}), (a2,b2)=({a2},{b2}) */",
    fmt("GLV_A1", m_a1, 2), f"#define GLV_A1_NEG {s_a1}",
    fmt("GLV_B1", m_b1, 2), f"#define GLV_B1_NEG {s_b1}",
    fmt("GLV_A2", m_a2, 2), f"#define GLV_A2_NEG {s_a2}",
    fmt("GLV_B2", m_b2, 2), f"#define GLV_B2_NEG {s_b2}",
    f"/* g_i = round(2^320 * (b2, -b1) / r): fixed-point multipliers */",
    fmt("GLV_G1", g1, 4),
    fmt("GLV_G2", g2, 4),
    "#define GLV_SHIFT 320",
    "",
    "#endif /* BN254_GLV_H */",
]
path = sys.argv[1] if len(sys.argv) > 1 else "include/bn254_glv.h"
open(path, "w").write("\n".join(lines) + "\n")
print(f"verified + wrote {path}")
print(f"|k1| max bits {maxk1.bit_length()}, |k2| max bits {maxk2.bit_length()}")
print(f"a1 bits {a1.bit_length()}, b1 {b1.bit_length()}, a2 {a2.bit_length()}, b2 {b2.bit_length()}")
