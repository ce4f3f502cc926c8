"""Pure-Python big-int restatement of the Pasta curves (Pallas / Vesta).

ORACLE TEST INFRASTRUCTURE — this module is part of the CPU oracle for the
Taiga/Halo2 hot path.  Only `tests/`, `__graft_entry__.smoke()` and
`bench.py`'s cpu_baseline leg may import anything under `oracle/`.  The
product path (taiga_amd / libtaiga_gpu.so) never imports or links this.

Restates the arithmetic of the `pasta_curves` crate v0.5.1 (the heliaxdev
fork used by /root/reference taiga_halo2/Cargo.toml:9; base crate zcash
pasta_curves), which is NOT vendored under /root/reference.
Constants below are the published Pasta parameters (printed in the
reference's book/src/spec.md context and in the public crate):

  p  (Fp = pallas::Base  = vesta::Scalar)
     = 0x40000000000000000000000000000000224698fc094cf91b992d30ed00000001
  q  (Fq = vesta::Base   = pallas::Scalar)
     = 0x40000000000000000000000000000000224698fc0994a8dd8c46eb2100000001
  Both curves: y^2 = x^3 + 5, generator (-1, 2).
  Fp and Fq both have 2-adicity S = 32 with multiplicative generator 5.

Conventions restated from the pasta_curves crate and PINNED in-container by
tests/test_srs_pin.py against the reference's bundled SRS
(/root/reference/taiga_halo2/params/params_15, committed as
tests/golden/params_15): point compression is x.to_repr() (32-byte
little-endian) with the sign bit (y odd) in bit 255; identity = all zeros;
g_lagrange[i] = n^-1 * sum_j omega^{-ij} g[j] with omega the order-2^k root
of unity of the SCALAR field derived from generator 5.
"""

P = 0x40000000000000000000000000000000224698FC094CF91B992D30ED00000001
Q = 0x40000000000000000000000000000000224698FC0994A8DD8C46EB2100000001
S = 32  # 2-adicity of both p-1 and q-1
MULT_GEN = 5  # multiplicative generator of both Fp* and Fq*
B = 5  # curve coefficient: y^2 = x^3 + 5


def root_of_unity(mod: int, k: int) -> int:
    """Order-2^k root of unity: MULT_GEN^((mod-1)/2^k) mod mod.

    Matches pasta_curves' ROOT_OF_UNITY (= 5^((p-1)/2^S)) raised to
    2^(S-k), as halo2's EvaluationDomain does (get_omega squaring chain).
    """
    assert k <= S
    return pow(MULT_GEN, (mod - 1) >> k, mod)


def sqrt_mod(a: int, mod: int):
    """Tonelli–Shanks for mod ≡ 1 (mod 2^32). Returns a root or None."""
    a %= mod
    if a == 0:
        return 0
    if pow(a, (mod - 1) // 2, mod) != 1:
        return None
    # mod - 1 = t * 2^S with t odd
    t = (mod - 1) >> S
    z = pow(MULT_GEN, t, mod)  # generator of the 2-Sylow subgroup
    m = S
    c = z
    u = pow(a, t, mod)
    r = pow(a, (t + 1) // 2, mod)
    while u != 1:
        # find least i with u^(2^i) = 1
        i = 0
        u2 = u
        while u2 != 1:
            u2 = u2 * u2 % mod
            i += 1
        b = pow(c, 1 << (m - i - 1), mod)
        m = i
        c = b * b % mod
        u = u * c % mod
        r = r * b % mod
    return r


class Point:
    """A point on y^2 = x^3 + 5 over F_mod, affine big-int coords.

    mod == Q → Vesta (the commitment curve of Params<vesta::Affine>;
    scalars in Fp);  mod == P → Pallas.
    """

    __slots__ = ("x", "y", "inf", "mod")

    def __init__(self, x, y, mod, inf=False):
        self.x = x % mod
        self.y = y % mod
        self.mod = mod
        self.inf = inf

    @classmethod
    def identity(cls, mod):
        return cls(0, 0, mod, True)

    @classmethod
    def generator(cls, mod):
        # pasta_curves: generator = (-1, 2) on both curves
        return cls(mod - 1, 2, mod)

    def is_on_curve(self):
        if self.inf:
            return True
        m = self.mod
        return (self.y * self.y - (self.x * self.x % m * self.x + B)) % m == 0

    def __eq__(self, o):
        if self.inf or o.inf:
            return self.inf and o.inf
        return self.x == o.x and self.y == o.y and self.mod == o.mod

    def __neg__(self):
        if self.inf:
            return self
        return Point(self.x, self.mod - self.y, self.mod)

    def __add__(self, o):
        if self.inf:
            return o
        if o.inf:
            return self
        m = self.mod
        if self.x == o.x:
            if (self.y + o.y) % m == 0:
                return Point.identity(m)
            return self.double()
        lam = (o.y - self.y) * pow(o.x - self.x, -1, m) % m
        x3 = (lam * lam - self.x - o.x) % m
        y3 = (lam * (self.x - x3) - self.y) % m
        return Point(x3, y3, m)

    def double(self):
        if self.inf:
            return self
        m = self.mod
        lam = 3 * self.x * self.x % m * pow(2 * self.y % m, -1, m) % m
        x3 = (lam * lam - 2 * self.x) % m
        y3 = (lam * (self.x - x3) - self.y) % m
        return Point(x3, y3, m)

    def mul(self, k: int):
        """Scalar multiple (k taken as plain integer; caller reduces)."""
        r = Point.identity(self.mod)
        a = self
        while k:
            if k & 1:
                r = r + a
            a = a.double()
            k >>= 1
        return r

    # --- pasta_curves GroupEncoding (32-byte compressed) ---
    def to_bytes(self) -> bytes:
        if self.inf:
            return b"\x00" * 32
        buf = bytearray(self.x.to_bytes(32, "little"))
        buf[31] |= (self.y & 1) << 7
        return bytes(buf)

    @classmethod
    def from_bytes(cls, b: bytes, mod: int):
        assert len(b) == 32
        buf = bytearray(b)
        sign = buf[31] >> 7
        buf[31] &= 0x7F
        x = int.from_bytes(bytes(buf), "little")
        if x == 0 and sign == 0 and all(v == 0 for v in buf):
            return cls.identity(mod)
        if x >= mod:
            return None
        y = sqrt_mod((x * x % mod * x + B) % mod, mod)
        if y is None:
            return None
        if (y & 1) != sign:
            y = mod - y
        return cls(x, y, mod)


def msm(scalars, points):
    """Naive-but-windowed multi-scalar multiplication (oracle scale only)."""
    if not points:
        return None
    mod = points[0].mod
    W = 8
    nwin = (256 + W - 1) // W
    acc = Point.identity(mod)
    for w in reversed(range(nwin)):
        for _ in range(W):
            acc = acc.double()
        buckets = [None] * (1 << W)
        for s, pt in zip(scalars, points):
            d = (s >> (w * W)) & ((1 << W) - 1)
            if d:
                buckets[d] = pt if buckets[d] is None else buckets[d] + pt
        run = Point.identity(mod)
        tot = Point.identity(mod)
        for d in reversed(range(1, 1 << W)):
            if buckets[d] is not None:
                run = run + buckets[d]
            tot = tot + run
        acc = acc + tot
    return acc


def ntt(values, omega, mod):
    """In-order radix-2 Cooley–Tukey NTT: out[i] = sum_j a[j] omega^{ij}.

    Matches halo2's best_fft semantics (bit-reverse + butterflies).
    Field-element version; used to cross-check the C oracle on small sizes.
    """
    n = len(values)
    a = list(values)
    logn = n.bit_length() - 1
    assert 1 << logn == n
    # bit-reverse permute
    for i in range(n):
        j = int(format(i, f"0{logn}b")[::-1], 2) if logn else 0
        if j > i:
            a[i], a[j] = a[j], a[i]
    size = 2
    while size <= n:
        wstep = pow(omega, n // size, mod)
        half = size // 2
        for start in range(0, n, size):
            w = 1
            for k in range(half):
                t = a[start + k + half] * w % mod
                a[start + k + half] = (a[start + k] - t) % mod
                a[start + k] = (a[start + k] + t) % mod
                w = w * wstep % mod
        size *= 2
    return a
