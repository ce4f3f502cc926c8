"""Blake2sChip (circuit/blake2s.rs, fully readable in-reference; restated).

Gates (blake2s.rs:244-364): field->8-word decompose, word->4-byte
decompose, byte->8-bit decompose, per-bit xor (3-row), u32 add with carry
bool, 4-word->field encode. The chip hashes 2-field messages with the
VPCommit personalization into a 256-bit digest exposed as two 16-byte-half
field elements (resource_logic_commitment_gadget).
"""
from ..plonkish import assign_free_advice
from .. import fields as F

IV = [0x6A09E667, 0xBB67AE85, 0x3C6EF372, 0xA54FF53A,
      0x510E527F, 0x9B05688C, 0x1F83D9AB, 0x5BE0CD19]

SIGMA = [
    [0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15],
    [14, 10, 4, 8, 9, 15, 13, 6, 1, 12, 0, 2, 11, 7, 5, 3],
    [11, 8, 12, 0, 5, 2, 15, 13, 10, 14, 3, 6, 7, 1, 9, 4],
    [7, 9, 3, 1, 13, 12, 11, 14, 2, 6, 5, 10, 4, 0, 15, 8],
    [9, 0, 5, 7, 2, 4, 10, 15, 14, 1, 11, 12, 6, 8, 3, 13],
    [2, 12, 6, 10, 0, 11, 8, 3, 4, 13, 7, 5, 15, 14, 1, 9],
    [12, 5, 1, 15, 14, 13, 4, 10, 0, 7, 6, 3, 9, 2, 8, 11],
    [13, 11, 7, 14, 12, 1, 3, 9, 5, 0, 15, 4, 8, 6, 2, 10],
    [6, 15, 14, 9, 11, 3, 0, 8, 12, 2, 13, 7, 1, 4, 10, 5],
    [10, 2, 8, 4, 7, 6, 1, 5, 15, 11, 9, 14, 3, 12, 13, 0],
]
R1, R2, R3, R4 = 16, 12, 8, 7
ROUNDS = 10


class Word:
    __slots__ = ("word", "bits")

    def __init__(self, word_cell, bit_cells):
        assert len(bit_cells) == 32
        self.word = word_cell
        self.bits = list(bit_cells)


class Blake2sConfig:
    def __init__(self, cs, advices):
        self.cs = cs
        self.adv = advices
        self.s_field = cs.selector()
        self.s_word = cs.selector()
        self.s_byte = cs.selector()
        self.s_xor = cs.selector()
        self.s_add = cs.selector()
        self.s_enc = cs.selector()
        a = advices

        # decompose field to 8 words (little-endian 32-bit limbs)
        words = [a[i].cur() for i in range(8)]
        fe = a[0].next()
        acc = words[0]
        for i in range(1, 8):
            acc = acc + words[i] * pow(2, 32 * i, F.P)
        cs.create_gate("decompose field to words", self.s_field,
                       [("field", acc - fe)])

        bytes_q = [a[i].cur() for i in range(4)]
        word_q = a[0].next()
        acc = bytes_q[0]
        for i in range(1, 4):
            acc = acc + bytes_q[i] * (1 << (8 * i))
        cs.create_gate("decompose word to bytes", self.s_word,
                       [("word", acc - word_q)])

        bits_q = [a[i].cur() for i in range(8)]
        byte_q = a[0].next()
        acc = bits_q[0]
        for i in range(1, 8):
            acc = acc + bits_q[i] * (1 << i)
        cs.create_gate("decompose byte to bits", self.s_byte,
                       [("byte", acc - byte_q)])

        xor_cons = []
        for i in range(8):
            lhs = a[i].prev()
            rhs = a[i].cur()
            out = a[i].next()
            xor_cons.append((f"bit{i}", lhs + rhs - lhs * rhs * 2 - out))
        cs.create_gate("byte xor", self.s_xor, xor_cons)

        lhs = a[0].cur()
        rhs = a[1].cur()
        out = a[0].next()
        carry = a[1].next()
        cs.create_gate("word add", self.s_add, [
            ("carry bool", carry * (carry - 1)),
            ("equal", lhs + rhs - carry * (1 << 32) - out),
        ])

        words4 = [a[i].cur() for i in range(4)]
        fe = a[0].next()
        acc = words4[0]
        for i in range(1, 4):
            acc = acc + words4[i] * pow(2, 32 * i, F.P)
        cs.create_gate("encode four words to one field", self.s_enc,
                       [("encode", acc - fe)])


class Blake2sChip:
    def __init__(self, config: Blake2sConfig):
        self.cfg = config
        self.cs = config.cs

    # --- byte/bit plumbing -------------------------------------------------
    def _byte_from_v(self, byte_v):
        """Blake2sByte::from_u8: 2-row region; returns (byte_cell, bit_cells)."""
        cfg = self.cfg
        with self.cs.region("byte decompose") as r:
            cfg.s_byte.enable(r, 0)
            bits = [r.assign_advice(cfg.adv[i], 0, byte_v.bit(i)) for i in range(8)]
            byte = r.assign_advice(cfg.adv[0], 1, byte_v)
        return byte, bits

    def _byte_from_const(self, value):
        cfg = self.cfg
        with self.cs.region("byte decompose const") as r:
            cfg.s_byte.enable(r, 0)
            bits = [r.assign_advice_from_constant(cfg.adv[i], 0, (value >> i) & 1)
                    for i in range(8)]
            byte = r.assign_advice_from_constant(cfg.adv[0], 1, value)
        return byte, bits

    def _word_decompose(self, byte_cells, word_cell):
        cfg = self.cfg
        with self.cs.region("word decompose") as r:
            cfg.s_word.enable(r, 0)
            for i, b in enumerate(byte_cells):
                r.copy_advice(b, cfg.adv[i], 0)
            r.copy_advice(word_cell, cfg.adv[0], 1)

    def _byte_decompose(self, bit_cells, byte_cell):
        cfg = self.cfg
        with self.cs.region("byte recompose") as r:
            cfg.s_byte.enable(r, 0)
            for i, b in enumerate(bit_cells):
                r.copy_advice(b, cfg.adv[i], 0)
            r.copy_advice(byte_cell, cfg.adv[0], 1)

    def word_from_const(self, value):
        cfg = self.cfg
        byte_cells = []
        bit_cells = []
        tmp = value
        for _ in range(4):
            byte, bits = self._byte_from_const(tmp & 0xFF)
            byte_cells.append(byte)
            bit_cells += bits
            tmp >>= 8
        from ..plonkish import assign_free_constant
        word = assign_free_constant(self.cs, cfg.adv[0], value)
        self._word_decompose(byte_cells, word)
        return Word(word, bit_cells)

    def from_word(self, word_cell):
        """Blake2sWord::from_word: decompose a u32-valued cell."""
        cfg = self.cfg
        byte_cells = []
        bit_cells = []
        for i in range(4):
            byte, bits = self._byte_from_v(word_cell.reg.byte(i))
            byte_cells.append(byte)
            bit_cells += bits
        self._word_decompose(byte_cells, word_cell)
        return Word(word_cell, bit_cells)

    def from_bits(self, bit_cells):
        cfg = self.cfg
        byte_cells = []
        for c in range(4):
            chunk = bit_cells[c * 8:(c + 1) * 8]
            bv = chunk[0].reg
            for i in range(1, 8):
                bv = bv + chunk[i].reg * (1 << i)
            byte = assign_free_advice(self.cs, cfg.adv[8], bv)
            self._byte_decompose(chunk, byte)
            byte_cells.append(byte)
        wv = byte_cells[0].reg
        for i in range(1, 4):
            wv = wv + byte_cells[i].reg * (1 << (8 * i))
        word = assign_free_advice(self.cs, cfg.adv[8], wv)
        self._word_decompose(byte_cells, word)
        return Word(word, bit_cells)

    def byte_xor(self, x_bits, y_bits):
        cfg = self.cfg
        with self.cs.region("byte xor") as r:
            cfg.s_xor.enable(r, 1)
            out = []
            for i in range(8):
                r.copy_advice(x_bits[i], cfg.adv[i], 0)
                r.copy_advice(y_bits[i], cfg.adv[i], 1)
                xv = x_bits[i].reg
                yv = y_bits[i].reg
                out.append(r.assign_advice(cfg.adv[i], 2, xv + yv - xv * yv * 2))
        return out

    def word_xor(self, x_bits, y_bits):
        bits = []
        for c in range(4):
            bits += self.byte_xor(x_bits[c * 8:(c + 1) * 8], y_bits[c * 8:(c + 1) * 8])
        return bits

    def add_mod_u32(self, x_cell, y_cell):
        cfg = self.cfg
        with self.cs.region("word add") as r:
            cfg.s_add.enable(r, 0)
            x = r.copy_advice(x_cell, cfg.adv[0], 0)
            y = r.copy_advice(y_cell, cfg.adv[1], 0)
            s = x.reg + y.reg
            carry = s.byte(4)
            ret = s - carry * (1 << 32)
            rc = r.assign_advice(cfg.adv[0], 1, ret)
            r.assign_advice(cfg.adv[1], 1, carry)
        return rc

    # --- compression -------------------------------------------------------
    def field_decompose(self, field_cell):
        cfg = self.cfg
        bits = []
        byte_cells = []
        for i in range(32):
            byte, bbits = self._byte_from_v(field_cell.reg.byte(i))
            bits += bbits
            byte_cells.append(byte)
        word_cells = []
        for c in range(8):
            chunk = byte_cells[c * 4:(c + 1) * 4]
            wv = chunk[0].reg
            for i in range(1, 4):
                wv = wv + chunk[i].reg * (1 << (8 * i))
            word = assign_free_advice(self.cs, cfg.adv[8], wv)
            self._word_decompose(chunk, word)
            word_cells.append(word)
        with self.cs.region("field decompose") as r:
            cfg.s_field.enable(r, 0)
            for i, w in enumerate(word_cells):
                r.copy_advice(w, cfg.adv[i], 0)
            r.copy_advice(field_cell, cfg.adv[0], 1)
        return [Word(word_cells[c], bits[c * 32:(c + 1) * 32]) for c in range(8)]

    def g(self, v, a, b, c, d, x, y):
        s1 = self.add_mod_u32(v[a].word, v[b].word)
        s2 = self.add_mod_u32(s1, x.word)
        v[a] = self.from_word(s2)

        bits = self.word_xor(v[d].bits, v[a].bits)
        v[d] = self.from_bits(bits[R1:] + bits[:R1])

        s = self.add_mod_u32(v[c].word, v[d].word)
        v[c] = self.from_word(s)

        bits = self.word_xor(v[b].bits, v[c].bits)
        v[b] = self.from_bits(bits[R2:] + bits[:R2])

        s1 = self.add_mod_u32(v[a].word, v[b].word)
        s2 = self.add_mod_u32(s1, y.word)
        v[a] = self.from_word(s2)

        bits = self.word_xor(v[d].bits, v[a].bits)
        v[d] = self.from_bits(bits[R3:] + bits[:R3])

        s = self.add_mod_u32(v[c].word, v[d].word)
        v[c] = self.from_word(s)

        bits = self.word_xor(v[b].bits, v[c].bits)
        v[b] = self.from_bits(bits[R4:] + bits[:R4])

    def compress(self, h, m, t, f):
        v = list(h)
        for iv in IV[0:4]:
            v.append(self.word_from_const(iv))
        v.append(self.word_from_const(IV[4] ^ (t & 0xFFFFFFFF)))
        v.append(self.word_from_const(IV[5] ^ ((t >> 32) & 0xFFFFFFFF)))
        v.append(self.word_from_const(IV[6] ^ (0xFFFFFFFF if f else 0)))
        v.append(self.word_from_const(IV[7]))
        assert len(v) == 16
        for rnd in range(ROUNDS):
            s = SIGMA[rnd % 10]
            self.g(v, 0, 4, 8, 12, m[s[0]], m[s[1]])
            self.g(v, 1, 5, 9, 13, m[s[2]], m[s[3]])
            self.g(v, 2, 6, 10, 14, m[s[4]], m[s[5]])
            self.g(v, 3, 7, 11, 15, m[s[6]], m[s[7]])
            self.g(v, 0, 5, 10, 15, m[s[8]], m[s[9]])
            self.g(v, 1, 6, 11, 12, m[s[10]], m[s[11]])
            self.g(v, 2, 7, 8, 13, m[s[12]], m[s[13]])
            self.g(v, 3, 4, 9, 14, m[s[14]], m[s[15]])
        for i in range(8):
            bits = self.word_xor(h[i].bits, v[i].bits)
            bits = self.word_xor(bits, v[i + 8].bits)
            h[i] = self.from_bits(bits)

    def process(self, input_cells, personalization: bytes):
        assert len(personalization) == 8
        assert len(input_cells) % 2 == 0
        h = [
            self.word_from_const(IV[0] ^ 0x01010000 ^ 32),
            self.word_from_const(IV[1]),
            self.word_from_const(IV[2]),
            self.word_from_const(IV[3]),
            self.word_from_const(IV[4]),
            self.word_from_const(IV[5]),
            self.word_from_const(IV[6] ^ int.from_bytes(personalization[0:4], "little")),
            self.word_from_const(IV[7] ^ int.from_bytes(personalization[4:8], "little")),
        ]
        blocks = []
        for i in range(0, len(input_cells), 2):
            block = []
            for fcell in input_cells[i:i + 2]:
                block += self.field_decompose(fcell)
            blocks.append(block)
        if not blocks:
            blocks.append([self.word_from_const(0) for _ in range(16)])
        for i, block in enumerate(blocks[:-1]):
            self.compress(h, block, (i + 1) * 64, False)
        self.compress(h, blocks[-1], len(blocks) * 64, True)
        return h

    def encode_result(self, h):
        cfg = self.cfg
        assert len(h) == 8
        fields = []
        for c in range(2):
            words = h[c * 4:(c + 1) * 4]
            with self.cs.region("encode words") as r:
                cfg.s_enc.enable(r, 0)
                for i, w in enumerate(words):
                    r.copy_advice(w.word, cfg.adv[i], 0)
                fv = words[0].word.reg
                for i in range(1, 4):
                    fv = fv + words[i].word.reg * pow(2, 32 * i, F.P)
                fields.append(r.assign_advice(cfg.adv[0], 1, fv))
        return fields


def resource_logic_commitment_gadget(chip: Blake2sChip, resource_logic_cell, rcm_cell):
    """blake2s.rs:23-35."""
    h = chip.process([resource_logic_cell, rcm_cell], b"VPCommit")
    return chip.encode_result(h)
