"""Minimal PNG encode/decode over zlib (capability analog of the reference's
libpng-backed core/lib/png; enough for tf.image.encode_png/decode_png and
tf.summary.image: 8-bit grayscale/RGB/RGBA, no interlace)."""
import struct
import zlib

import numpy as np

_SIG = b'\x89PNG\r\n\x1a\n'


def _chunk(tag, data):
    out = struct.pack('>I', len(data)) + tag + data
    crc = zlib.crc32(tag + data) & 0xFFFFFFFF
    return out + struct.pack('>I', crc)


def encode_png(arr, compression=6):
    """arr: uint8 [h, w] or [h, w, c] with c in 1/2/3/4."""
    arr = np.ascontiguousarray(arr, dtype=np.uint8)
    if arr.ndim == 2:
        arr = arr[:, :, None]
    h, w, c = arr.shape
    color_type = {1: 0, 2: 4, 3: 2, 4: 6}[c]
    ihdr = struct.pack('>IIBBBBB', w, h, 8, color_type, 0, 0, 0)
    # filter byte 0 (None) per scanline
    raw = b''.join(b'\x00' + arr[y].tobytes() for y in range(h))
    idat = zlib.compress(raw, compression)
    return (_SIG + _chunk(b'IHDR', ihdr) + _chunk(b'IDAT', idat) +
            _chunk(b'IEND', b''))


def _unfilter(data, h, w, c):
    stride = w * c
    out = np.zeros((h, stride), dtype=np.uint8)
    pos = 0
    prev = np.zeros(stride, dtype=np.uint8)
    for y in range(h):
        ft = data[pos]
        pos += 1
        line = np.frombuffer(data[pos:pos + stride],
                             dtype=np.uint8).astype(np.int32)
        pos += stride
        if ft == 0:
            cur = line
        elif ft == 1:  # Sub
            cur = line.copy()
            for i in range(c, stride):
                cur[i] = (cur[i] + cur[i - c]) & 0xFF
        elif ft == 2:  # Up
            cur = (line + prev) & 0xFF
        elif ft == 3:  # Average
            cur = line.copy()
            for i in range(stride):
                left = cur[i - c] if i >= c else 0
                cur[i] = (cur[i] + ((left + int(prev[i])) >> 1)) & 0xFF
        elif ft == 4:  # Paeth
            cur = line.copy()
            for i in range(stride):
                a = int(cur[i - c]) if i >= c else 0
                b = int(prev[i])
                cc = int(prev[i - c]) if i >= c else 0
                p = a + b - cc
                pa, pb, pc = abs(p - a), abs(p - b), abs(p - cc)
                pred = a if (pa <= pb and pa <= pc) else \
                    (b if pb <= pc else cc)
                cur[i] = (cur[i] + pred) & 0xFF
        else:
            raise ValueError('bad PNG filter %d' % ft)
        out[y] = cur.astype(np.uint8)
        prev = out[y]
    return out.reshape(h, w, c)


def decode_png(data):
    if data[:8] != _SIG:
        raise ValueError('not a PNG')
    pos = 8
    idat = b''
    w = h = bit_depth = color_type = None
    while pos < len(data):
        (length,) = struct.unpack('>I', data[pos:pos + 4])
        tag = data[pos + 4:pos + 8]
        body = data[pos + 8:pos + 8 + length]
        pos += 12 + length
        if tag == b'IHDR':
            w, h, bit_depth, color_type, _, _, interlace = struct.unpack(
                '>IIBBBBB', body)
            if bit_depth != 8 or interlace:
                raise ValueError('only 8-bit non-interlaced PNG supported')
        elif tag == b'IDAT':
            idat += body
        elif tag == b'IEND':
            break
    c = {0: 1, 2: 3, 4: 2, 6: 4}[color_type]
    raw = zlib.decompress(idat)
    return _unfilter(raw, h, w, c)
