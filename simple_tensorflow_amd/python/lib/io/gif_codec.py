"""GIF87a/89a decoder in pure python (capability analog of the reference's
core/lib/gif/gif_io.cc, which wraps giflib): LZW decompression, local/global
color tables, interlacing, multi-frame compositing with disposal methods,
transparency. decode_gif returns uint8 [num_frames, h, w, 3] like the
reference's DecodeGif."""
import struct

import numpy as np


def _read_color_table(data, off, size):
    table = np.frombuffer(data[off:off + 3 * size], np.uint8)
    return table.reshape(size, 3), off + 3 * size


def _lzw_decode(min_code_size, data):
    """Standard GIF LZW: variable-width codes, clear/EOI, dict of byte
    strings."""
    clear = 1 << min_code_size
    eoi = clear + 1
    out = bytearray()

    def reset():
        d = {i: bytes([i]) for i in range(clear)}
        return d, clear + 2, min_code_size + 1

    table, next_code, width = reset()
    prev = None
    acc = 0
    nbits = 0
    for byte in data:
        acc |= byte << nbits
        nbits += 8
        while nbits >= width:
            code = acc & ((1 << width) - 1)
            acc >>= width
            nbits -= width
            if code == clear:
                table, next_code, width = reset()
                prev = None
                continue
            if code == eoi:
                return bytes(out)
            if prev is None:
                entry = table[code]
            elif code in table:
                entry = table[code]
            elif code == next_code:
                entry = prev + prev[:1]
            else:
                raise ValueError('gif: bad LZW code')
            out += entry
            if prev is not None and next_code < 4096:
                table[next_code] = prev + entry[:1]
                next_code += 1
                # dictionary-style decoders add entries one code later than
                # the encoder does, so the width must grow one entry early
                if next_code == (1 << width) - 1 and width < 12:
                    width += 1
            prev = entry
    return bytes(out)


def _deinterlace(indices, w, h):
    out = np.empty((h, w), np.uint8)
    rows = indices.reshape(h, w)
    passes = [(0, 8), (4, 8), (2, 4), (1, 2)]
    src = 0
    for start, step in passes:
        for y in range(start, h, step):
            out[y] = rows[src]
            src += 1
    return out


def decode_gif(data):
    data = bytes(data)
    if data[:6] not in (b'GIF87a', b'GIF89a'):
        raise ValueError('gif: bad signature')
    W, H, flags, bg, _ = struct.unpack('<HHBBB', data[6:13])
    off = 13
    gct = None
    if flags & 0x80:
        gct, off = _read_color_table(data, off, 2 << (flags & 7))
    frames = []
    canvas = np.zeros((H, W, 3), np.uint8)
    if gct is not None:
        canvas[:] = gct[bg]
    transparent = None
    disposal = 0
    prev_canvas = None
    while off < len(data):
        block = data[off]
        off += 1
        if block == 0x3B:  # trailer
            break
        if block == 0x21:  # extension
            label = data[off]
            off += 1
            if label == 0xF9:  # graphics control
                size = data[off]
                gflags, _delay, tindex = struct.unpack(
                    '<BHB', data[off + 1:off + 5])
                off += 1 + size
                transparent = tindex if gflags & 1 else None
                disposal = (gflags >> 2) & 7
            # skip remaining sub-blocks
            while True:
                sz = data[off]
                off += 1
                if sz == 0:
                    break
                off += sz
        elif block == 0x2C:  # image descriptor
            x, y, w, h, iflags = struct.unpack('<HHHHB', data[off:off + 9])
            off += 9
            ct = gct
            if iflags & 0x80:
                ct, off = _read_color_table(data, off, 2 << (iflags & 7))
            if ct is None:
                raise ValueError('gif: no color table')
            min_code = data[off]
            off += 1
            lzw = bytearray()
            while True:
                sz = data[off]
                off += 1
                if sz == 0:
                    break
                lzw += data[off:off + sz]
                off += sz
            idx = np.frombuffer(_lzw_decode(min_code, bytes(lzw)),
                                np.uint8)[:w * h]
            if iflags & 0x40:
                idx = _deinterlace(idx, w, h)
            else:
                idx = idx.reshape(h, w)
            if disposal == 3 and prev_canvas is not None:
                base = prev_canvas.copy()
            else:
                base = canvas.copy()
            prev_canvas = canvas.copy()
            region = ct[idx]
            if transparent is not None:
                mask = idx != transparent
                sub = base[y:y + h, x:x + w]
                sub[mask] = region[mask]
                base[y:y + h, x:x + w] = sub
            else:
                base[y:y + h, x:x + w] = region
            frames.append(base)
            if disposal == 2:  # restore background
                canvas = np.zeros((H, W, 3), np.uint8)
                if gct is not None:
                    canvas[:] = gct[bg]
            elif disposal == 3:
                canvas = prev_canvas.copy()
            else:
                canvas = base.copy()
        else:
            raise ValueError('gif: unknown block 0x%02x' % block)
    if not frames:
        raise ValueError('gif: no frames')
    return np.stack(frames)


def encode_gif(frames, loop=0):
    """Minimal GIF89a encoder (for round-trip tests): one global 256-color
    table built from the first frame via uniform 6x7x6 quantization."""
    frames = np.asarray(frames, np.uint8)
    if frames.ndim == 3:
        frames = frames[None]
    n, h, w, _ = frames.shape
    # 6*7*6=252-entry uniform palette
    rs, gs, bs = np.linspace(0, 255, 6), np.linspace(0, 255, 7), \
        np.linspace(0, 255, 6)
    palette = np.zeros((256, 3), np.uint8)
    i = 0
    for r in rs:
        for g in gs:
            for b in bs:
                palette[i] = (int(r), int(g), int(b))
                i += 1

    def quant(img):
        ri = np.clip((img[..., 0].astype(int) * 6) // 256, 0, 5)
        gi = np.clip((img[..., 1].astype(int) * 7) // 256, 0, 6)
        bi = np.clip((img[..., 2].astype(int) * 6) // 256, 0, 5)
        return (ri * 42 + gi * 6 + bi).astype(np.uint8)

    out = bytearray(b'GIF89a')
    out += struct.pack('<HHBBB', w, h, 0x80 | 7, 0, 0)
    out += palette.tobytes()
    for f in range(n):
        out += b'\x2c' + struct.pack('<HHHHB', 0, 0, w, h, 0)
        idx = quant(frames[f]).reshape(-1)
        out += bytes([8])  # min code size
        out += _lzw_encode(8, idx.tobytes())
        out += b'\x00'
    out += b'\x3b'
    return bytes(out)


def _lzw_encode(min_code_size, data):
    clear = 1 << min_code_size
    eoi = clear + 1
    codes = []
    table = {bytes([i]): i for i in range(clear)}
    next_code = clear + 2
    width = min_code_size + 1
    codes.append((clear, width))
    cur = b''
    for byte in data:
        nxt = cur + bytes([byte])
        if nxt in table:
            cur = nxt
        else:
            codes.append((table[cur], width))
            if next_code < 4096:
                table[nxt] = next_code
                next_code += 1
                # mirror the decoder: width grows once the NEXT assignable
                # code no longer fits
                if next_code == (1 << width) and width < 12:
                    width += 1
            else:
                codes.append((clear, width))
                table = {bytes([i]): i for i in range(clear)}
                next_code = clear + 2
                width = min_code_size + 1
            cur = bytes([byte])
    if cur:
        codes.append((table[cur], width))
    codes.append((eoi, width))
    acc = 0
    nbits = 0
    payload = bytearray()
    for code, cwidth in codes:
        acc |= code << nbits
        nbits += cwidth
        while nbits >= 8:
            payload.append(acc & 0xFF)
            acc >>= 8
            nbits -= 8
    if nbits:
        payload.append(acc & 0xFF)
    out = bytearray()
    for i in range(0, len(payload), 255):
        chunk = payload[i:i + 255]
        out += bytes([len(chunk)]) + chunk
    return bytes(out)
