"""LookupRangeCheckConfig<pallas::Base, 10> (halo2_gadgets 0.3
utilities/lookup_range_check.rs, un-vendored; restated).

One lookup argument into the 2^10 table column:
  input = q_lookup * ( q_running * (z_cur - z_next * 2^K)
                     + (q_lookup - q_running) * z_cur )
(q_lookup, q_running are complex selectors -> own fixed columns), plus the
bitshift gate for short checks:
  q_bitshift * (word(prev) * 2^(K - num_bits) - shifted(cur))
with 2^(K-num_bits) itself constrained via the constants column
(assign_advice_from_constant at cur+1 — kept as halo2 does to preserve the
advice query at Rotation::next on the running-sum column).
"""
from ..plonkish import ConstraintSystem
from .. import fields as F

K = 10


class LookupRangeCheckConfig:
    def __init__(self, cs: ConstraintSystem, running_sum, table_idx):
        self.cs = cs
        self.running_sum = running_sum
        self.table_idx = table_idx
        cs.enable_equality(running_sum)
        self.q_lookup = cs.complex_selector()
        self.q_running = cs.complex_selector()
        self.q_bitshift = cs.selector()

        z_cur = running_sum.cur()
        z_next = running_sum.next()
        ql = self.q_lookup.expr()
        qr = self.q_running.expr()
        running_word = z_cur - z_next * (1 << K)
        inp = ql * (qr * running_word + (ql - qr) * z_cur)
        cs.add_lookup("range check", [inp], [table_idx.cur()])

        word = running_sum.prev()
        shifted = running_sum.cur()
        two_pow = running_sum.next()
        cs.create_gate("bitshift", self.q_bitshift,
                       [("shifted", word * two_pow - shifted)])

    def witness_check(self, value_v, num_words, strict=False):
        """witness_check: decompose value into num_words K-bit words via a
        running sum; returns zs[0..num_words] cells (zs[0] = value)."""
        cs = self.cs
        inv2k = pow(1 << K, F.P - 2, F.P)
        with cs.region("range check") as r:
            zs = [r.assign_advice(self.running_sum, 0, value_v)]
            z = value_v
            for i in range(num_words):
                self.q_lookup.enable(r, i)
                self.q_running.enable(r, i)
                # k_i = z mod 2^K (via bit extraction of the canonical repr)
                k_word = z.bit(0)
                for b in range(1, K):
                    k_word = k_word + z.bit(b) * (1 << b)
                z = (z - k_word) * inv2k
                zs.append(r.assign_advice(self.running_sum, i + 1, z))
            if strict:
                raise NotImplementedError("strict unused by the compliance path")
        return zs

    def copy_short_check(self, cell, num_bits):
        """copy_short_check: element < 2^num_bits via shifted lookup."""
        cs = self.cs
        with cs.region(f"short range check {num_bits}") as r:
            word = r.copy_advice(cell, self.running_sum, 0)
            # the word itself is ALSO looked up (without it, word*2^(K-n)
            # mod p could alias back into the table — unsound)
            self.q_lookup.enable(r, 0)
            self.q_lookup.enable(r, 1)
            self.q_bitshift.enable(r, 1)
            shifted = word.reg * (1 << (K - num_bits))
            r.assign_advice(self.running_sum, 1, shifted)
            r.assign_advice_from_constant(self.running_sum, 2, 1 << (K - num_bits))
