"""Independent pure-Python model of the PAX microblock format.

Third implementation (next to the C oracle and the HIP engine), mirroring the
reference's unit-test strategy of checking decoders against independently
computed expectations (test_column_decoder.h:65-185 — the reference ships no
stored golden vectors, so the pin is N-version agreement + format fidelity).

Implements, from the struct definitions cited in oracle/obx_format.h:
  header/col-header parse, bit-stream reads (little-endian bit order),
  RAW / DICT / RLE / CONST / INTEGER_BASE_DIFF row decode, NULL extraction.
All values are returned as Python ints (already sign-extended) or None (NULL).
"""
import struct

ENC_RAW, ENC_DICT, ENC_RLE, ENC_CONST, ENC_INT_DIFF, ENC_SDIFF, ENC_HEX = range(7)
ENC_SPREFIX = 7
ENC_COLEQ = 8
ENC_SUBSTR = 9

ATTR_FIX = 1
ATTR_EXT = 2
ATTR_BP = 4

SC_INT, SC_STRING, SC_DECIMAL = 1, 5, 4


def store_class(obj_type):
    return {4: SC_INT, 5: SC_INT, 19: SC_INT, 23: SC_STRING,
            50: SC_DECIMAL}[obj_type]


def type_store_size(obj_type):
    return {4: 4, 5: 8, 19: 4, 23: -1, 50: -1}[obj_type]


def crc32c(buf):
    """CRC-32C (Castagnoli, reflected 0x82f63b78, init 0, no final xor) —
    the value ob_crc64_sse42 stores in data_checksum (zero-extended to i64);
    independent table-free implementation from the polynomial alone."""
    crc = 0
    for b in buf:
        crc ^= b
        for _ in range(8):
            crc = (0x82F63B78 ^ (crc >> 1)) if crc & 1 else crc >> 1
    return crc


def bs_get(buf, pos, length):
    """bit read, little-endian bit order (ob_bit_stream.h semantics)."""
    v = 0
    for i in range(length):
        p = pos + i
        if (buf[p >> 3] >> (p & 7)) & 1:
            v |= 1 << i
    return v


def sign_extend(v, nbytes):
    bits = nbytes * 8
    if nbytes < 8 and (v >> (bits - 1)) & 1:
        v -= 1 << bits
    elif nbytes == 8 and (v >> 63) & 1:
        v -= 1 << 64
    return v


class Block:
    def __init__(self, data, schema):
        """schema: list of (obj_type, scale, precision, len)."""
        self.data = data
        self.schema = schema
        (self.magic, self.version, self.header_size, self.header_checksum,
         self.column_count, self.rowkey_column_count, self.flag16,
         self.row_count, self.row_store_type, self.opt,
         self.var_column_count, self.row_data_offset) = struct.unpack_from(
            "<hhIhHHHIBBHI", data, 0)
        assert self.magic == 1005, self.magic
        assert self.version == 3
        assert self.column_count == len(schema)
        (self.data_zlength, self.data_checksum) = struct.unpack_from(
            "<iq", data, 44)  # after original_length(4)+mmtv(8)+data_length(4)
        assert self.data_checksum == crc32c(data[64:self.data_zlength]), \
            "data_checksum mismatch (payload CRC-32C)"
        self.row_index_byte = self.opt & 7
        self.extend_value_bit = (self.opt >> 3) & 7
        self.col_headers = []
        off = self.header_size
        for _ in range(self.column_count):
            ver, typ, attr, obj_type, evo, coff, clen = struct.unpack_from(
                "<bbbBIII", data, off)
            self.col_headers.append(dict(version=ver, type=typ, attr=attr,
                                         obj_type=obj_type, offset=coff,
                                         length=clen))
            off += 16
        self.meta_base = off  # header + col headers

    def decode_col(self, c):
        """Returns list of per-row values: int (sign-extended for ObIntSC,
        raw bytes as little-endian int for char) or None for NULL."""
        ch = self.col_headers[c]
        obj_type, _, _, dlen = self.schema[c]
        assert obj_type == ch["obj_type"]
        sc = store_class(obj_type)
        tss = type_store_size(obj_type)
        d = self.data
        base = self.meta_base + ch["offset"]
        rows = self.row_count
        evb = self.extend_value_bit
        has_ext = bool(ch["attr"] & ATTR_EXT)
        bp = bool(ch["attr"] & ATTR_BP)
        out = []
        t = ch["type"]
        if t == ENC_RAW:
            ext_bits = evb * rows if has_ext else 0
            if bp:
                k = ch["length"]
                for r in range(rows):
                    if has_ext and bs_get(d, base + 0 + r * evb * 8 * 0, 0):
                        pass
                    if has_ext and bs_get(d[base:], r * evb, evb):
                        out.append(None)
                        continue
                    v = bs_get(d[base:], ext_bits + r * k, k)
                    out.append(v)  # bit-packed: zero-extended
            else:
                fl = ch["length"]
                fix0 = base + (ext_bits + 7) // 8
                for r in range(rows):
                    if has_ext and bs_get(d[base:], r * evb, evb):
                        out.append(None)
                        continue
                    v = int.from_bytes(d[fix0 + r * fl:fix0 + (r + 1) * fl],
                                       "little")
                    if sc == SC_INT:
                        v = sign_extend(v, tss)
                    elif sc == SC_DECIMAL:
                        v = sign_extend(v, fl)
                    out.append(v)
        elif t == ENC_DICT:
            ver, ref_size, count, data_size, attr = struct.unpack_from(
                "<BBIHB", d, base)
            pay = base + 9
            refs = base + ch["length"]
            for r in range(rows):
                if bp:
                    ref = bs_get(d[refs:], r * ref_size, ref_size)
                else:
                    ref = int.from_bytes(
                        d[refs + r * ref_size:refs + (r + 1) * ref_size],
                        "little")
                if ref >= count:
                    out.append(None)
                    continue
                v = int.from_bytes(
                    d[pay + ref * data_size:pay + (ref + 1) * data_size],
                    "little")
                if sc == SC_INT:
                    v = sign_extend(v, tss)
                elif sc == SC_DECIMAL:
                    v = sign_extend(v, data_size)
                out.append(v)
        elif t == ENC_RLE:
            ver, attr, count, doff = struct.unpack_from("<BBII", d, base)
            rib, rfb = attr & 7, (attr >> 3) & 7
            rid0 = base + 10
            ref0 = rid0 + count * rib
            starts = [int.from_bytes(d[rid0 + i * rib:rid0 + (i + 1) * rib],
                                     "little") for i in range(count)]
            refs = [int.from_bytes(d[ref0 + i * rfb:ref0 + (i + 1) * rfb],
                                   "little") for i in range(count)]
            dm = base + doff
            dver, drs, dcount, dsize, dattr = struct.unpack_from("<BBIHB", d, dm)
            pay = dm + 9
            run = 0
            for r in range(rows):
                while run + 1 < count and starts[run + 1] <= r:
                    run += 1
                ref = refs[run]
                if ref >= dcount:
                    out.append(None)
                    continue
                v = int.from_bytes(d[pay + ref * dsize:pay + (ref + 1) * dsize],
                                   "little")
                if sc == SC_INT:
                    v = sign_extend(v, tss)
                elif sc == SC_DECIMAL:
                    v = sign_extend(v, dsize)
                out.append(v)
        elif t == ENC_CONST:
            ver, cnt, const_ref, attr, off2 = struct.unpack_from("<BBBBH", d,
                                                                 base)
            rib = attr & 7
            if cnt == 0:
                if const_ref > 0:
                    return [None] * rows
                cell = tss if sc == SC_INT else dlen
                v = int.from_bytes(d[base + off2:base + off2 + cell], "little")
                if sc == SC_INT:
                    v = sign_extend(v, tss)
                elif sc == SC_DECIMAL:
                    v = sign_extend(v, cell)
                return [v] * rows
            exc_ref0 = base + 6
            exc_rid0 = exc_ref0 + cnt
            exc = {}
            for i in range(cnt):
                rid = int.from_bytes(
                    d[exc_rid0 + i * rib:exc_rid0 + (i + 1) * rib], "little")
                exc[rid] = d[exc_ref0 + i]
            dm = base + off2
            dver, drs, dcount, dsize, dattr = struct.unpack_from("<BBIHB", d, dm)
            pay = dm + 9
            for r in range(rows):
                ref = exc.get(r, const_ref)
                if ref >= dcount:
                    out.append(None)
                    continue
                v = int.from_bytes(d[pay + ref * dsize:pay + (ref + 1) * dsize],
                                   "little")
                if sc == SC_INT:
                    v = sign_extend(v, tss)
                elif sc == SC_DECIMAL:
                    v = sign_extend(v, dsize)
                out.append(v)
        elif t == ENC_INT_DIFF:
            ver, k = struct.unpack_from("<BB", d, base)
            bval = int.from_bytes(d[base + 2:base + 2 + tss], "little")
            bval = sign_extend(bval, tss) if sc == SC_INT else bval
            data0 = base + ch["length"]
            ext_bits = evb * rows if has_ext else 0
            for r in range(rows):
                if has_ext and bs_get(d[data0:], r * evb, evb):
                    out.append(None)
                    continue
                if bp:
                    diff = bs_get(d[data0:], ext_bits + r * k, k)
                else:
                    f0 = data0 + (ext_bits + 7) // 8
                    diff = int.from_bytes(d[f0 + r * k:f0 + (r + 1) * k],
                                          "little")
                s = (bval + diff) & ((1 << 64) - 1)  # mod-2^64 wrap
                out.append(sign_extend(s, 8) if sc == SC_INT else s)
        elif t == ENC_SDIFF:
            # obx_sdiff_meta: version u8, hex_char_cnt u8, string_size u16,
            # diff_desc_cnt u8; then descs, hex chars, common bytes
            ver, hex_cnt, ssize, nd = struct.unpack_from("<BBHB", d, base)
            descs = list(d[base + 5:base + 5 + nd])
            chars = list(d[base + 5 + nd:base + 5 + nd + hex_cnt])
            common = d[base + 5 + nd + hex_cnt:base + ch["length"]]
            diff_len = sum(dd >> 1 for dd in descs if dd & 1)
            stride = (diff_len + 1) // 2 if hex_cnt else diff_len
            data0 = base + ch["length"]
            ext_bits = evb * rows if has_ext else 0
            fix0 = data0 + (ext_bits + 7) // 8
            for r in range(rows):
                if has_ext and bs_get(d[data0:], r * evb, evb):
                    out.append(None)
                    continue
                rp = d[fix0 + r * stride:fix0 + (r + 1) * stride]
                v = 0
                pos = cpos = dpos = 0
                for dd in descs:
                    cnt = dd >> 1
                    if dd & 1:
                        for _ in range(cnt):
                            if hex_cnt:
                                nib = (rp[dpos // 2] >>
                                       (((dpos + 1) % 2) * 4)) & 0xF
                                cc = chars[nib]
                            else:
                                cc = rp[dpos]
                            v |= cc << (8 * pos)
                            pos += 1
                            dpos += 1
                    else:
                        for _ in range(cnt):
                            v |= common[cpos] << (8 * pos)
                            pos += 1
                            cpos += 1
                out.append(v)
        elif t == ENC_HEX:
            # obx_hex_meta: version u8, char_cnt u8, string_size u16; chars
            ver, nch, ssize = struct.unpack_from("<BBH", d, base)
            chars = list(d[base + 4:base + 4 + nch])
            stride = (ssize + 1) // 2
            data0 = base + ch["length"]
            ext_bits = evb * rows if has_ext else 0
            fix0 = data0 + (ext_bits + 7) // 8
            for r in range(rows):
                if has_ext and bs_get(d[data0:], r * evb, evb):
                    out.append(None)
                    continue
                rp = d[fix0 + r * stride:fix0 + (r + 1) * stride]
                v = 0
                for i in range(ssize):
                    nib = (rp[i // 2] >> (((i + 1) % 2) * 4)) & 0xF
                    v |= chars[nib] << (8 * i)
                out.append(v)
        elif t == ENC_SPREFIX:
            # obx_sprefix_meta: version u8, count u8, string_size u16,
            # hex_char_cnt u8, pib u8; then hex chars, END offsets, prefixes
            ver, pcnt, ssize, hex_cnt, pib = struct.unpack_from(
                "<BBHBB", d, base)
            p0 = base + 6
            chars = list(d[p0:p0 + hex_cnt])
            e0 = p0 + hex_cnt
            ends = [int.from_bytes(d[e0 + j * pib:e0 + (j + 1) * pib],
                                   "little") for j in range(pcnt)]
            pdata = e0 + pcnt * pib
            plens = [ends[0]] + [ends[j] - ends[j - 1]
                                 for j in range(1, pcnt)]
            max_suffix = ssize - min(plens)
            stride = 1 + ((max_suffix + 1) // 2 if hex_cnt else max_suffix)
            data0 = base + ch["length"]
            ext_bits = evb * rows if has_ext else 0
            fix0 = data0 + (ext_bits + 7) // 8
            for r in range(rows):
                if has_ext and bs_get(d[data0:], r * evb, evb):
                    out.append(None)
                    continue
                rp = d[fix0 + r * stride:fix0 + (r + 1) * stride]
                ref = rp[0] & 0xF
                pstart = ends[ref - 1] if ref else 0
                plen = ends[ref] - pstart
                v = 0
                for i in range(plen):
                    v |= d[pdata + pstart + i] << (8 * i)
                for i in range(ssize - plen):
                    if hex_cnt:
                        nib = (rp[1 + i // 2] >>
                               (((i + 1) % 2) * 4)) & 0xF
                        cc = chars[nib]
                    else:
                        cc = rp[1 + i]
                    v |= cc << (8 * (plen + i))
                out.append(v)
        elif t == ENC_COLEQ:
            # obx_coleq_meta: version u8, ref_col u16, exc_cnt u16, rib u8
            ver, ref_col, exc_cnt, rib = struct.unpack_from("<BHHB", d, base)
            p0 = base + 6
            rids = [int.from_bytes(d[p0 + i * rib:p0 + (i + 1) * rib],
                                   "little") for i in range(exc_cnt)]
            nb0 = p0 + exc_cnt * rib
            datp = nb0 + (exc_cnt + 7) // 8
            ref_vals = self.decode_col(ref_col)
            out = list(ref_vals)
            for i, rid in enumerate(rids):
                if (d[nb0 + i // 8] >> (i % 8)) & 1:
                    out[rid] = None
                else:
                    v = int.from_bytes(d[datp + i * dlen:
                                         datp + (i + 1) * dlen], "little")
                    if sc == SC_INT:
                        v = sign_extend(v, tss)
                    elif sc == SC_DECIMAL:
                        v = sign_extend(v, dlen)
                    out[rid] = v
        elif t == ENC_SUBSTR:
            # obx_substr_meta: version u8, attr u8, start u16, ref u16,
            # exc_cnt u16, rib u8, ref_len u8
            (ver, attr, start, ref_col, exc_cnt, rib,
             ref_len) = struct.unpack_from("<BBHHHBB", d, base)
            p0 = base + 10
            rids = [int.from_bytes(d[p0 + i * rib:p0 + (i + 1) * rib],
                                   "little") for i in range(exc_cnt)]
            nb0 = p0 + exc_cnt * rib
            datp = nb0 + (exc_cnt + 7) // 8
            ref_vals = self.decode_col(ref_col)
            mask = (1 << (8 * dlen)) - 1
            out = [None if v is None else (v >> (8 * start)) & mask
                   for v in ref_vals]
            for i, rid in enumerate(rids):
                if (d[nb0 + i // 8] >> (i % 8)) & 1:
                    out[rid] = None
                else:
                    out[rid] = int.from_bytes(
                        d[datp + i * dlen:datp + (i + 1) * dlen], "little")
        else:
            raise NotImplementedError(t)
        return out


def eval_leaf(op, v, lo, hi, in_list, sc, length):
    """white-op semantics; v None = NULL."""
    if op == 8:   # NU
        return v is None
    if op == 9:   # NN
        return v is not None
    if v is None:
        return False

    def key(x):
        if sc == SC_STRING:
            return int.from_bytes(
                int(x % (1 << 64)).to_bytes(8, "little")[:length], "big")
        return x
    x, l, h = key(v), key(lo), key(hi)
    return {0: x == l, 1: x <= l, 2: x < l, 3: x >= l, 4: x > l,
            5: x != l, 6: l <= x <= h,
            7: x in [key(e) for e in in_list]}[op]
