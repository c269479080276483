"""TFRecord file format: length-delimited records with masked crc32c framing
(byte-compatible with the reference's lib/io/record_writer.cc /
record_reader.cc; the same framing carries TFEvents files)."""
import struct

_CRC_TABLE = []


def _make_table():
    poly = 0x82F63B78
    for i in range(256):
        c = i
        for _ in range(8):
            c = (poly ^ (c >> 1)) if (c & 1) else (c >> 1)
        _CRC_TABLE.append(c)


_make_table()


def crc32c(data):
    crc = 0xFFFFFFFF
    for b in data:
        crc = _CRC_TABLE[(crc ^ b) & 0xFF] ^ (crc >> 8)
    return crc ^ 0xFFFFFFFF


def masked_crc32c(data):
    crc = crc32c(data)
    return (((crc >> 15) | (crc << 17)) + 0xA282EAD8) & 0xFFFFFFFF


class TFRecordWriter(object):
    def __init__(self, path):
        self._f = open(path, 'wb')

    def write(self, record):
        if isinstance(record, str):
            record = record.encode()
        header = struct.pack('<Q', len(record))
        self._f.write(header)
        self._f.write(struct.pack('<I', masked_crc32c(header)))
        self._f.write(record)
        self._f.write(struct.pack('<I', masked_crc32c(record)))

    def flush(self):
        self._f.flush()

    def close(self):
        self._f.close()

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


def tf_record_iterator(path, options=None):
    with open(path, 'rb') as f:
        while True:
            header = f.read(8)
            if len(header) < 8:
                return
            (length,) = struct.unpack('<Q', header)
            f.read(4)  # header crc
            data = f.read(length)
            if len(data) < length:
                return
            f.read(4)  # data crc
            yield data
