"""Pure-Python BN254 reference model (plain bignum arithmetic).

This is the independent ground truth the C++ oracle is validated against
(the Rust reference prover cannot be built here, so the
oracle is pinned by (a) this model, (b) the in-repo structural pins, and
(c) published keccak256 vectors).  Deliberately naive and slow — use only
at tiny sizes.
"""

Q = 21888242871839275222246405745257275088696311157297823662689037894645226208583
R = 21888242871839275222246405745257275088548364400416034343698204186575808495617
R256 = 1 << 256

FR_GEN = 5
FR_TWO_ADICITY = 28


def to_mont(x, p):
    return (x * R256) % p


def from_mont(x, p):
    return (x * pow(R256, -1, p)) % p


def limbs_to_int(limbs):
    v = 0
    for i, l in enumerate(limbs):
        v |= int(l) << (64 * i)
    return v


def int_to_limbs(v, n=4):
    return [(v >> (64 * i)) & 0xFFFFFFFFFFFFFFFF for i in range(n)]


def fr_root_of_unity(n):
    t = (R - 1) >> FR_TWO_ADICITY
    root = pow(FR_GEN, t, R)
    logn = n.bit_length() - 1
    assert 1 << logn == n
    for _ in range(FR_TWO_ADICITY - logn):
        root = root * root % R
    return root


def ntt_direct(a, inverse=False):
    """O(n^2) DFT over Fr. a: list of ints mod R. Natural order both ways."""
    n = len(a)
    w = fr_root_of_unity(n)
    if inverse:
        w = pow(w, -1, R)
    out = []
    for k in range(n):
        s = 0
        wk = pow(w, k, R)
        x = 1
        for j in range(n):
            s = (s + a[j] * x) % R
            x = x * wk % R
    # note: x runs w^(jk)
        out.append(s)
    if inverse:
        ninv = pow(n, -1, R)
        out = [v * ninv % R for v in out]
    return out


# ---- EC over Fq: y^2 = x^3 + 3, affine with None = infinity ----

def ec_add(p1, p2):
    if p1 is None:
        return p2
    if p2 is None:
        return p1
    x1, y1 = p1
    x2, y2 = p2
    if x1 == x2:
        if (y1 + y2) % Q == 0:
            return None
        lam = (3 * x1 * x1) * pow(2 * y1, -1, Q) % Q
    else:
        lam = (y2 - y1) * pow(x2 - x1, -1, Q) % Q
    x3 = (lam * lam - x1 - x2) % Q
    y3 = (lam * (x1 - x3) - y1) % Q
    return (x3, y3)


def ec_mul(p, k):
    acc = None
    while k:
        if k & 1:
            acc = ec_add(acc, p)
        p = ec_add(p, p)
        k >>= 1
    return acc


def ec_is_on_curve(p):
    if p is None:
        return True
    x, y = p
    return (y * y - x * x * x - 3) % Q == 0


G1_GEN = (1, 2)


def msm_ref(points, scalars):
    acc = None
    for p, s in zip(points, scalars):
        acc = ec_add(acc, ec_mul(p, s))
    return acc
