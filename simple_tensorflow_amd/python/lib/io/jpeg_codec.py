"""Baseline JFIF/JPEG codec in pure python+numpy (capability analog of the
reference's core/lib/jpeg/jpeg_mem.cc, which wraps libjpeg).

Encoder: baseline sequential, 4:4:4, standard Annex-K quantization tables
scaled by quality, standard huffman tables. Decoder: baseline sequential
with 4:4:4 / 4:2:0 / 4:2:2 chroma subsampling, restart markers, grayscale.
"""
import struct

import numpy as np

# Annex K luminance / chrominance quantization tables (zigzag order applied
# at use time; stored row-major here).
_QY = np.array([
    16, 11, 10, 16, 24, 40, 51, 61,
    12, 12, 14, 19, 26, 58, 60, 55,
    14, 13, 16, 24, 40, 57, 69, 56,
    14, 17, 22, 29, 51, 87, 80, 62,
    18, 22, 37, 56, 68, 109, 103, 77,
    24, 35, 55, 64, 81, 104, 113, 92,
    49, 64, 78, 87, 103, 121, 120, 101,
    72, 92, 95, 98, 112, 100, 103, 99], np.float64).reshape(8, 8)
_QC = np.array([
    17, 18, 24, 47, 99, 99, 99, 99,
    18, 21, 26, 66, 99, 99, 99, 99,
    24, 26, 56, 99, 99, 99, 99, 99,
    47, 66, 99, 99, 99, 99, 99, 99,
    99, 99, 99, 99, 99, 99, 99, 99,
    99, 99, 99, 99, 99, 99, 99, 99,
    99, 99, 99, 99, 99, 99, 99, 99,
    99, 99, 99, 99, 99, 99, 99, 99], np.float64).reshape(8, 8)

_ZIGZAG = np.array([
    0, 1, 8, 16, 9, 2, 3, 10, 17, 24, 32, 25, 18, 11, 4, 5,
    12, 19, 26, 33, 40, 48, 41, 34, 27, 20, 13, 6, 7, 14, 21, 28,
    35, 42, 49, 56, 57, 50, 43, 36, 29, 22, 15, 23, 30, 37, 44, 51,
    58, 59, 52, 45, 38, 31, 39, 46, 53, 60, 61, 54, 47, 55, 62, 63])

# standard huffman tables (JPEG Annex K): (bits[1..16], values)
_HT_DC_LUM = ([0, 1, 5, 1, 1, 1, 1, 1, 1, 0, 0, 0, 0, 0, 0, 0],
              list(range(12)))
_HT_DC_CHR = ([0, 3, 1, 1, 1, 1, 1, 1, 1, 1, 1, 0, 0, 0, 0, 0],
              list(range(12)))
_HT_AC_LUM = ([0, 2, 1, 3, 3, 2, 4, 3, 5, 5, 4, 4, 0, 0, 1, 0x7d], [
    0x01, 0x02, 0x03, 0x00, 0x04, 0x11, 0x05, 0x12, 0x21, 0x31, 0x41, 0x06,
    0x13, 0x51, 0x61, 0x07, 0x22, 0x71, 0x14, 0x32, 0x81, 0x91, 0xa1, 0x08,
    0x23, 0x42, 0xb1, 0xc1, 0x15, 0x52, 0xd1, 0xf0, 0x24, 0x33, 0x62, 0x72,
    0x82, 0x09, 0x0a, 0x16, 0x17, 0x18, 0x19, 0x1a, 0x25, 0x26, 0x27, 0x28,
    0x29, 0x2a, 0x34, 0x35, 0x36, 0x37, 0x38, 0x39, 0x3a, 0x43, 0x44, 0x45,
    0x46, 0x47, 0x48, 0x49, 0x4a, 0x53, 0x54, 0x55, 0x56, 0x57, 0x58, 0x59,
    0x5a, 0x63, 0x64, 0x65, 0x66, 0x67, 0x68, 0x69, 0x6a, 0x73, 0x74, 0x75,
    0x76, 0x77, 0x78, 0x79, 0x7a, 0x83, 0x84, 0x85, 0x86, 0x87, 0x88, 0x89,
    0x8a, 0x92, 0x93, 0x94, 0x95, 0x96, 0x97, 0x98, 0x99, 0x9a, 0xa2, 0xa3,
    0xa4, 0xa5, 0xa6, 0xa7, 0xa8, 0xa9, 0xaa, 0xb2, 0xb3, 0xb4, 0xb5, 0xb6,
    0xb7, 0xb8, 0xb9, 0xba, 0xc2, 0xc3, 0xc4, 0xc5, 0xc6, 0xc7, 0xc8, 0xc9,
    0xca, 0xd2, 0xd3, 0xd4, 0xd5, 0xd6, 0xd7, 0xd8, 0xd9, 0xda, 0xe1, 0xe2,
    0xe3, 0xe4, 0xe5, 0xe6, 0xe7, 0xe8, 0xe9, 0xea, 0xf1, 0xf2, 0xf3, 0xf4,
    0xf5, 0xf6, 0xf7, 0xf8, 0xf9, 0xfa])
_HT_AC_CHR = ([0, 2, 1, 2, 4, 4, 3, 4, 7, 5, 4, 4, 0, 1, 2, 0x77], [
    0x00, 0x01, 0x02, 0x03, 0x11, 0x04, 0x05, 0x21, 0x31, 0x06, 0x12, 0x41,
    0x51, 0x07, 0x61, 0x71, 0x13, 0x22, 0x32, 0x81, 0x08, 0x14, 0x42, 0x91,
    0xa1, 0xb1, 0xc1, 0x09, 0x23, 0x33, 0x52, 0xf0, 0x15, 0x62, 0x72, 0xd1,
    0x0a, 0x16, 0x24, 0x34, 0xe1, 0x25, 0xf1, 0x17, 0x18, 0x19, 0x1a, 0x26,
    0x27, 0x28, 0x29, 0x2a, 0x35, 0x36, 0x37, 0x38, 0x39, 0x3a, 0x43, 0x44,
    0x45, 0x46, 0x47, 0x48, 0x49, 0x4a, 0x53, 0x54, 0x55, 0x56, 0x57, 0x58,
    0x59, 0x5a, 0x63, 0x64, 0x65, 0x66, 0x67, 0x68, 0x69, 0x6a, 0x73, 0x74,
    0x75, 0x76, 0x77, 0x78, 0x79, 0x7a, 0x82, 0x83, 0x84, 0x85, 0x86, 0x87,
    0x88, 0x89, 0x8a, 0x92, 0x93, 0x94, 0x95, 0x96, 0x97, 0x98, 0x99, 0x9a,
    0xa2, 0xa3, 0xa4, 0xa5, 0xa6, 0xa7, 0xa8, 0xa9, 0xaa, 0xb2, 0xb3, 0xb4,
    0xb5, 0xb6, 0xb7, 0xb8, 0xb9, 0xba, 0xc2, 0xc3, 0xc4, 0xc5, 0xc6, 0xc7,
    0xc8, 0xc9, 0xca, 0xd2, 0xd3, 0xd4, 0xd5, 0xd6, 0xd7, 0xd8, 0xd9, 0xda,
    0xe2, 0xe3, 0xe4, 0xe5, 0xe6, 0xe7, 0xe8, 0xe9, 0xea, 0xf2, 0xf3, 0xf4,
    0xf5, 0xf6, 0xf7, 0xf8, 0xf9, 0xfa])


def _dct_matrix():
    m = np.zeros((8, 8))
    for k in range(8):
        for n in range(8):
            m[k, n] = np.cos(np.pi * (2 * n + 1) * k / 16.0)
    m *= 0.5
    m[0, :] *= 1.0 / np.sqrt(2.0)
    return m


_DCT = _dct_matrix()


def _build_codes(bits, values):
    """(bits, values) -> {symbol: (code, length)}"""
    codes = {}
    code = 0
    k = 0
    for length in range(1, 17):
        for _ in range(bits[length - 1]):
            codes[values[k]] = (code, length)
            code += 1
            k += 1
        code <<= 1
    return codes


def _build_decoder(bits, values):
    """(bits, values) -> {(length, code): symbol}"""
    table = {}
    code = 0
    k = 0
    for length in range(1, 17):
        for _ in range(bits[length - 1]):
            table[(length, code)] = values[k]
            code += 1
            k += 1
        code <<= 1
    return table


class _BitWriter(object):
    def __init__(self):
        self.out = bytearray()
        self.acc = 0
        self.nbits = 0

    def put(self, code, length):
        self.acc = (self.acc << length) | (code & ((1 << length) - 1))
        self.nbits += length
        while self.nbits >= 8:
            byte = (self.acc >> (self.nbits - 8)) & 0xFF
            self.out.append(byte)
            if byte == 0xFF:
                self.out.append(0x00)  # byte stuffing
            self.nbits -= 8

    def flush(self):
        if self.nbits:
            pad = 8 - self.nbits
            self.put((1 << pad) - 1, pad)


def _magnitude(v):
    """(size, amplitude bits) for a DC diff / AC coefficient."""
    v = int(v)
    if v == 0:
        return 0, 0
    a = abs(v)
    size = a.bit_length()
    bits = v if v > 0 else v + (1 << size) - 1
    return size, bits


def encode_jpeg(arr, quality=75):
    """uint8 [h, w, 3] (or [h, w, 1]/[h, w]) -> baseline JFIF bytes, 4:4:4."""
    arr = np.asarray(arr, np.uint8)
    if arr.ndim == 2:
        arr = arr[:, :, None]
    h, w, c = arr.shape
    gray = c == 1
    if gray:
        y = arr[..., 0].astype(np.float64)
        cb = cr = None
    else:
        rgb = arr.astype(np.float64)
        y = 0.299 * rgb[..., 0] + 0.587 * rgb[..., 1] + 0.114 * rgb[..., 2]
        cb = -0.168736 * rgb[..., 0] - 0.331264 * rgb[..., 1] + 0.5 * rgb[..., 2] + 128
        cr = 0.5 * rgb[..., 0] - 0.418688 * rgb[..., 1] - 0.081312 * rgb[..., 2] + 128

    q = max(1, min(100, int(quality)))
    scale = 5000.0 / q if q < 50 else 200.0 - 2 * q
    qy = np.clip(np.floor((_QY * scale + 50) / 100), 1, 255)
    qc = np.clip(np.floor((_QC * scale + 50) / 100), 1, 255)

    dc_lum = _build_codes(*_HT_DC_LUM)
    ac_lum = _build_codes(*_HT_AC_LUM)
    dc_chr = _build_codes(*_HT_DC_CHR)
    ac_chr = _build_codes(*_HT_AC_CHR)

    bw = _BitWriter()
    prev_dc = [0, 0, 0]
    bh, bwid = (h + 7) // 8, (w + 7) // 8

    def pad(plane):
        return np.pad(plane, ((0, bh * 8 - h), (0, bwid * 8 - w)),
                      mode='edge')

    if gray:
        planes = [pad(y)]
        qtabs_l = [qy]
        dctabs = [dc_lum]
        actabs = [ac_lum]
    else:
        planes = [pad(y), pad(cb), pad(cr)]
        qtabs_l = [qy, qc, qc]
        dctabs = [dc_lum, dc_chr, dc_chr]
        actabs = [ac_lum, ac_chr, ac_chr]
    ncomp = len(planes)
    for by in range(bh):
        for bx in range(bwid):
            for comp in range(ncomp):
                block = planes[comp][by * 8:by * 8 + 8,
                                     bx * 8:bx * 8 + 8] - 128.0
                coeff = _DCT @ block @ _DCT.T
                quant = np.round(coeff / qtabs_l[comp]).astype(np.int64)
                zz = quant.reshape(-1)[_ZIGZAG]
                # DC
                diff = int(zz[0]) - prev_dc[comp]
                prev_dc[comp] = int(zz[0])
                size, bits = _magnitude(diff)
                code, ln = dctabs[comp][size]
                bw.put(code, ln)
                if size:
                    bw.put(bits, size)
                # AC
                run = 0
                for k in range(1, 64):
                    v = int(zz[k])
                    if v == 0:
                        run += 1
                        continue
                    while run > 15:
                        code, ln = actabs[comp][0xF0]
                        bw.put(code, ln)
                        run -= 16
                    size, bits = _magnitude(v)
                    code, ln = actabs[comp][(run << 4) | size]
                    bw.put(code, ln)
                    bw.put(bits, size)
                    run = 0
                if run:
                    code, ln = actabs[comp][0x00]  # EOB
                    bw.put(code, ln)
    bw.flush()

    def seg(marker, payload):
        return struct.pack('>BBH', 0xFF, marker, len(payload) + 2) + payload

    def dqt(tid, table):
        zz = table.reshape(-1)[_ZIGZAG].astype(np.uint8)
        return seg(0xDB, bytes([tid]) + zz.tobytes())

    def dht(cls, tid, bits, values):
        return seg(0xC4, bytes([(cls << 4) | tid] + bits + list(values)))

    out = b'\xff\xd8'  # SOI
    out += seg(0xE0, b'JFIF\x00\x01\x01\x00\x00\x01\x00\x01\x00\x00')
    out += dqt(0, qy)
    if not gray:
        out += dqt(1, qc)
    sof = struct.pack('>BHHB', 8, h, w, ncomp)
    specs = [(0x11, 0)] if gray else [(0x11, 0), (0x11, 1), (0x11, 1)]
    for cid, (sf, tq) in enumerate(specs, 1):
        sof += bytes([cid, sf, tq])
    out += seg(0xC0, sof)
    out += dht(0, 0, *_HT_DC_LUM) + dht(1, 0, *_HT_AC_LUM)
    if not gray:
        out += dht(0, 1, *_HT_DC_CHR) + dht(1, 1, *_HT_AC_CHR)
    if gray:
        sos = bytes([1, 1, 0x00, 0, 63, 0])
    else:
        sos = bytes([3, 1, 0x00, 2, 0x11, 3, 0x11, 0, 63, 0])
    out += seg(0xDA, sos)
    out += bytes(bw.out)
    out += b'\xff\xd9'  # EOI
    return out


class _BitReader(object):
    def __init__(self, data):
        self.data = data
        self.pos = 0
        self.acc = 0
        self.nbits = 0

    def _fill(self):
        while self.nbits <= 24 and self.pos < len(self.data):
            b = self.data[self.pos]
            self.pos += 1
            if b == 0xFF:
                nxt = self.data[self.pos] if self.pos < len(self.data) else 0
                if nxt == 0x00:
                    self.pos += 1  # stuffed
                elif 0xD0 <= nxt <= 0xD7:
                    # restart marker handled by caller via sync()
                    self.pos -= 1
                    b = None
                else:
                    self.pos -= 1
                    b = None
            if b is None:
                break
            self.acc = (self.acc << 8) | b
            self.nbits += 8

    def get(self, n):
        if n == 0:
            return 0
        self._fill()
        if self.nbits < n:
            # pad with zeros at stream end
            self.acc <<= (n - self.nbits)
            self.nbits = n
        v = (self.acc >> (self.nbits - n)) & ((1 << n) - 1)
        self.nbits -= n
        self.acc &= (1 << self.nbits) - 1
        return v

    def sync_restart(self):
        # drop partial byte, expect FFD0-FFD7
        self.acc = 0
        self.nbits = 0
        while self.pos + 1 < len(self.data):
            if self.data[self.pos] == 0xFF and \
                    0xD0 <= self.data[self.pos + 1] <= 0xD7:
                self.pos += 2
                return
            self.pos += 1


def _decode_huff(br, table):
    code = 0
    for length in range(1, 17):
        code = (code << 1) | br.get(1)
        sym = table.get((length, code))
        if sym is not None:
            return sym
    raise ValueError('jpeg: bad huffman code')


def _extend(bits, size):
    if size == 0:
        return 0
    if bits < (1 << (size - 1)):
        return bits - (1 << size) + 1
    return bits


def decode_jpeg(data):
    """Baseline JFIF bytes -> uint8 [h, w, channels] numpy array."""
    data = bytes(data)
    if data[:2] != b'\xff\xd8':
        raise ValueError('jpeg: missing SOI')
    pos = 2
    qtabs = {}
    huff = {}
    frame = None
    restart_interval = 0
    while pos < len(data):
        if data[pos] != 0xFF:
            raise ValueError('jpeg: bad marker')
        marker = data[pos + 1]
        pos += 2
        if marker == 0xD9:
            break
        length = struct.unpack('>H', data[pos:pos + 2])[0]
        payload = data[pos + 2:pos + length]
        if marker == 0xDB:  # DQT
            p = 0
            while p < len(payload):
                pq, tq = payload[p] >> 4, payload[p] & 15
                p += 1
                if pq:
                    t = np.frombuffer(payload[p:p + 128], '>u2').astype(
                        np.float64)
                    p += 128
                else:
                    t = np.frombuffer(payload[p:p + 64], np.uint8).astype(
                        np.float64)
                    p += 64
                tab = np.zeros(64)
                tab[_ZIGZAG] = t
                qtabs[tq] = tab.reshape(8, 8)
        elif marker in (0xC0, 0xC1):  # SOF0/1 baseline
            prec, h, w, nc = struct.unpack('>BHHB', payload[:6])
            comps = []
            for i in range(nc):
                cid, sf, tq = payload[6 + i * 3:9 + i * 3]
                comps.append({'id': cid, 'h': sf >> 4, 'v': sf & 15,
                              'tq': tq})
            frame = {'h': h, 'w': w, 'comps': comps}
        elif marker == 0xC4:  # DHT
            p = 0
            while p < len(payload):
                cls_id = payload[p]
                bits = list(payload[p + 1:p + 17])
                n = sum(bits)
                values = list(payload[p + 17:p + 17 + n])
                huff[(cls_id >> 4, cls_id & 15)] = _build_decoder(bits,
                                                                  values)
                p += 17 + n
        elif marker == 0xDD:  # DRI
            restart_interval = struct.unpack('>H', payload[:2])[0]
        elif marker == 0xDA:  # SOS
            ns = payload[0]
            scan = []
            for i in range(ns):
                cs, tt = payload[1 + i * 2], payload[2 + i * 2]
                scan.append({'cs': cs, 'dc': tt >> 4, 'ac': tt & 15})
            pos += length
            return _decode_scan(data, pos, frame, scan, qtabs, huff,
                                restart_interval)
        pos += length
    raise ValueError('jpeg: no scan data')


def _decode_scan(data, pos, frame, scan, qtabs, huff, restart_interval):
    h, w = frame['h'], frame['w']
    comps = frame['comps']
    hmax = max(c['h'] for c in comps)
    vmax = max(c['v'] for c in comps)
    mcux = (w + 8 * hmax - 1) // (8 * hmax)
    mcuy = (h + 8 * vmax - 1) // (8 * vmax)
    planes = []
    for c in comps:
        planes.append(np.zeros((mcuy * c['v'] * 8, mcux * c['h'] * 8)))
    scan_by_cs = {s['cs']: s for s in scan}
    br = _BitReader(data[:_find_eoi(data, pos)])
    br.pos = pos
    prev_dc = {c['id']: 0 for c in comps}
    mcu_count = 0
    for my in range(mcuy):
        for mx in range(mcux):
            if restart_interval and mcu_count and \
                    mcu_count % restart_interval == 0:
                br.sync_restart()
                for cid in prev_dc:
                    prev_dc[cid] = 0
            mcu_count += 1
            for ci, c in enumerate(comps):
                s = scan_by_cs[c['id']]
                dct = huff[(0, s['dc'])]
                act = huff[(1, s['ac'])]
                q = qtabs[c['tq']]
                for by in range(c['v']):
                    for bx in range(c['h']):
                        zz = np.zeros(64)
                        size = _decode_huff(br, dct)
                        diff = _extend(br.get(size), size)
                        prev_dc[c['id']] += diff
                        zz[0] = prev_dc[c['id']]
                        k = 1
                        while k < 64:
                            rs = _decode_huff(br, act)
                            r, sz = rs >> 4, rs & 15
                            if sz == 0:
                                if r == 15:
                                    k += 16
                                    continue
                                break  # EOB
                            k += r
                            if k > 63:
                                break
                            zz[k] = _extend(br.get(sz), sz)
                            k += 1
                        block = np.zeros(64)
                        block[_ZIGZAG] = zz
                        coeff = block.reshape(8, 8) * q
                        pix = _DCT.T @ coeff @ _DCT + 128.0
                        py0 = (my * c['v'] + by) * 8
                        px0 = (mx * c['h'] + bx) * 8
                        planes[ci][py0:py0 + 8, px0:px0 + 8] = pix
    # upsample to full res
    full = []
    for ci, c in enumerate(comps):
        p = planes[ci]
        ry, rx = vmax // c['v'], hmax // c['h']
        if ry > 1 or rx > 1:
            p = np.repeat(np.repeat(p, ry, 0), rx, 1)
        full.append(p[:h, :w])
    if len(comps) == 1:
        return np.clip(full[0], 0, 255).astype(np.uint8)[:, :, None]
    y, cb, cr = full[0], full[1] - 128.0, full[2] - 128.0
    r = y + 1.402 * cr
    g = y - 0.344136 * cb - 0.714136 * cr
    b = y + 1.772 * cb
    return np.clip(np.stack([r, g, b], -1), 0, 255).astype(np.uint8)


def _find_eoi(data, start):
    i = len(data) - 2
    while i > start:
        if data[i] == 0xFF and data[i + 1] == 0xD9:
            return i
        i -= 1
    return len(data)
