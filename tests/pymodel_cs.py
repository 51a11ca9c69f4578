"""Independent pure-Python model of the CS block format (RAW-codec
subset) — the CS analogue of pymodel.py: decodes obx_cs_block_enc
output from the BYTES alone, sharing no code with oracle/obx_cs*.c, so
format drift between the two implementations is caught. Codec-typed
streams (DZR/PFoR families) are pinned separately by hand-computed
byte vectors in test_cs_stream.py; this model covers the block walk,
headers, MSB-first bitmaps, null-replace recovery, dict/const-ref
layouts and string pooling with RAW streams.

Layout (cited in oracle/obx_cs_block.h):
  [16B obx header][12B all-col header][4B col header x n]
  [per-column data][pooled string bytes][stream-offset int stream]
"""
import struct

WB = [1, 2, 4, 8]


def _vi64(d, pos):
    u, sh, n = 0, 0, 0
    while True:
        b = d[pos + n]
        u |= (b & 0x7F) << sh
        n += 1
        if not (b & 0x80):
            break
        sh += 7
    u &= (1 << 64) - 1
    if u >= 1 << 63:
        u -= 1 << 64
    return u, pos + n


def _int_meta(d, pos):
    version, attr, typ, wtag = d[pos], d[pos + 1], d[pos + 2], d[pos + 3]
    pos += 4
    base = nrep = 0
    prec = None
    if attr & 0x1:
        base, pos = _vi64(d, pos)
    if attr & 0x2:
        nrep, pos = _vi64(d, pos)
    if attr & 0x4:
        prec = d[pos]
        pos += 1
    if version > 0:
        pos += 1  # pfor packing type
    return dict(attr=attr, typ=typ, wtag=wtag, base=base, nrep=nrep,
                prec=prec), pos


def _int_stream(d, pos, rows):
    """RAW integer stream -> signed values list."""
    m, pos = _int_meta(d, pos)
    assert m["typ"] == 1, "python model covers RAW streams only"
    wb = WB[m["wtag"]]
    vals = []
    for _ in range(rows):
        u = int.from_bytes(d[pos:pos + wb], "little")
        v = (u + m["base"]) & ((1 << 64) - 1)
        if v >= 1 << 63:
            v -= 1 << 64
        vals.append(v)
        pos += wb
    return m, vals, pos


def _str_meta(d, pos):
    version, attr = d[pos], d[pos + 1]
    pos += 2
    ul, pos = _vi64(d, pos)  # vi32 shares the byte format for >=0
    fl = 0
    if attr & 0x2:
        fl, pos = _vi64(d, pos)
    return dict(attr=attr, ul=ul, fl=fl), pos


def _bm(d, r):
    return (d[r // 8] >> (7 - r % 8)) & 1  # MSB-first


class CSBlock:
    def __init__(self, blob):
        (magic, version, hdr_sz, rows, ncols,
         _res) = struct.unpack_from("<IHHIHH", blob, 0)
        assert magic == 0x5343424F and version == 1
        self.rows, self.ncols = rows, ncols
        (_av, _aa, strlen, solen,
         scount) = struct.unpack_from("<BBIIH", blob, hdr_sz)
        ch0 = hdr_sz + 12
        self.cols = [struct.unpack_from("<BBBB", blob, ch0 + 4 * c)
                     for c in range(ncols)]
        # block-tail stream offsets (an int stream of scount values)
        so_start = len(blob) - solen
        self.soff = []
        if scount:
            _, self.soff, _ = _int_stream(blob, so_start, scount)
        self.pool = blob[so_start - strlen:so_start]
        self.blob = blob
        self._walk(ch0 + 4 * ncols)

    def _walk(self, pos):
        d, rows = self.blob, self.rows
        bmsz = (rows + 7) // 8
        si = 0
        str_off = 0
        self.decoded = []
        for c in range(self.ncols):
            _v, typ, attrs, _ot = self.cols[c]
            bitmap = None
            if attrs & 0x2:
                bitmap = d[pos:pos + bmsz]
                pos += bmsz
            if typ == 0:  # INTEGER
                m, vals, pos = _int_stream(d, pos, rows)
                assert pos == self.soff[si]
                si += 1
                nulls = set()
                if m["attr"] & 0x2:
                    nulls = {r for r in range(rows)
                             if vals[r] == m["nrep"]}
                elif bitmap is not None:
                    nulls = {r for r in range(rows) if _bm(bitmap, r)}
                self.decoded.append((
                    [None if r in nulls else vals[r]
                     for r in range(rows)]))
            elif typ == 1:  # STRING
                sm, pos = _str_meta(d, pos)
                assert pos == self.soff[si]
                si += 1
                data = self.pool[str_off:str_off + sm["ul"]]
                str_off += sm["ul"]
                if sm["attr"] & 0x2:  # fixed
                    fl = sm["fl"]
                    out = []
                    for r in range(rows):
                        if bitmap is not None and _bm(bitmap, r):
                            out.append(None)
                        else:
                            out.append(data[r * fl:(r + 1) * fl])
                else:
                    _, ends, pos = _int_stream(d, pos, rows)
                    assert pos == self.soff[si]
                    si += 1
                    out, prev = [], 0
                    for r in range(rows):
                        s = data[prev:ends[r]]
                        prev = ends[r]
                        isn = (_bm(bitmap, r) if bitmap is not None
                               else (sm["attr"] & 0x1 and len(s) == 0))
                        out.append(None if isn else s)
                self.decoded.append(out)
            elif typ in (2, 3):  # INT_DICT / STR_DICT
                (dver, dattrs, distinct,
                 ref_cnt) = struct.unpack_from("<BBII", d, pos)
                pos += 10
                if distinct == 0:
                    self.decoded.append([None] * rows)
                    continue
                if typ == 2:
                    _, dv, pos = _int_stream(d, pos, distinct)
                    assert pos == self.soff[si]
                    si += 1
                else:
                    sm, pos = _str_meta(d, pos)
                    assert pos == self.soff[si]
                    si += 1
                    ddata = self.pool[str_off:str_off + sm["ul"]]
                    str_off += sm["ul"]
                    if sm["attr"] & 0x2:
                        fl = sm["fl"]
                        dv = [ddata[i * fl:(i + 1) * fl]
                              for i in range(distinct)]
                    else:
                        _, dends, pos = _int_stream(d, pos, distinct)
                        assert pos == self.soff[si]
                        si += 1
                        dv, prev = [], 0
                        for e in dends:
                            dv.append(ddata[prev:e])
                            prev = e
                # ref stream
                _, ra, pos = _int_stream(d, pos, ref_cnt)
                assert pos == self.soff[si]
                si += 1
                if dattrs & 0x4:  # CONST_ENCODING_REF
                    ec, cref = ra[0], ra[1]
                    refs = [cref] * rows
                    for i in range(ec):
                        refs[ra[2 + i]] = ra[2 + ec + i]
                else:
                    refs = ra
                out = []
                for r in range(rows):
                    ref = refs[r]
                    if ref == distinct and (dattrs & 0x2):
                        out.append(None)
                    else:
                        out.append(dv[ref])
                self.decoded.append(out)
            else:
                raise AssertionError(typ)
