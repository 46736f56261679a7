"""Minimal MoPaQ (MPQ) archive reader — enough to pull
`replay.gamemetadata.json` out of a .SC2Replay for version routing.

The reference uses the `mpyq` package (`replay_decoder.py:361-380`); this
image ships no mpyq, so the decoder needs a self-contained reader.  Format
implemented from the public MPQ specification: user-data block, archive
header, encrypted hash/block tables (standard crypt table, seed 0x00100001),
single-unit and sectored files, zlib (0x02) and bzip2 (0x10) compression.
Encrypted *file data* is not supported (SC2 replay members are unencrypted).
"""
import bz2
import struct
import zlib

MPQ_USER_DATA_MAGIC = b'MPQ\x1b'
MPQ_HEADER_MAGIC = b'MPQ\x1a'

FLAG_EXISTS = 0x80000000
FLAG_ENCRYPTED = 0x00010000
FLAG_SINGLE_UNIT = 0x01000000
FLAG_COMPRESS = 0x00000200
FLAG_IMPLODE = 0x00000100

COMP_ZLIB = 0x02
COMP_BZIP2 = 0x10


def _build_crypt_table():
    table = [0] * 0x500
    seed = 0x00100001
    for i in range(0x100):
        index = i
        for j in range(5):
            seed = (seed * 125 + 3) % 0x2AAAAB
            t1 = (seed & 0xFFFF) << 0x10
            seed = (seed * 125 + 3) % 0x2AAAAB
            t2 = seed & 0xFFFF
            table[index] = t1 | t2
            index += 0x100
    return table

_CRYPT = _build_crypt_table()

HASH_TABLE_OFFSET = 0
HASH_NAME_A = 1
HASH_NAME_B = 2
HASH_FILE_KEY = 3


def mpq_hash(s: str, hash_type: int) -> int:
    seed1 = 0x7FED7FED
    seed2 = 0xEEEEEEEE
    for ch in s.upper():
        c = ord(ch)
        seed1 = (_CRYPT[(hash_type << 8) + c] ^ (seed1 + seed2)) & 0xFFFFFFFF
        seed2 = (c + seed1 + seed2 + (seed2 << 5) + 3) & 0xFFFFFFFF
    return seed1


def decrypt(data: bytes, key: int) -> bytes:
    seed1 = key & 0xFFFFFFFF
    seed2 = 0xEEEEEEEE
    out = bytearray()
    for i in range(len(data) // 4):
        seed2 = (seed2 + _CRYPT[0x400 + (seed1 & 0xFF)]) & 0xFFFFFFFF
        (value,) = struct.unpack_from('<I', data, i * 4)
        value = (value ^ (seed1 + seed2)) & 0xFFFFFFFF
        seed1 = ((~seed1 << 0x15) + 0x11111111 | seed1 >> 0x0B) & 0xFFFFFFFF
        seed2 = (value + seed2 + (seed2 << 5) + 3) & 0xFFFFFFFF
        out += struct.pack('<I', value)
    return bytes(out)


def encrypt(data: bytes, key: int) -> bytes:
    """Inverse of decrypt (used by tests to build synthetic archives)."""
    seed1 = key & 0xFFFFFFFF
    seed2 = 0xEEEEEEEE
    out = bytearray()
    for i in range(len(data) // 4):
        seed2 = (seed2 + _CRYPT[0x400 + (seed1 & 0xFF)]) & 0xFFFFFFFF
        (value,) = struct.unpack_from('<I', data, i * 4)
        enc = (value ^ (seed1 + seed2)) & 0xFFFFFFFF
        seed1 = ((~seed1 << 0x15) + 0x11111111 | seed1 >> 0x0B) & 0xFFFFFFFF
        seed2 = (value + seed2 + (seed2 << 5) + 3) & 0xFFFFFFFF
        out += struct.pack('<I', enc)
    return bytes(out)


def _decompress(data: bytes) -> bytes:
    method, payload = data[0], data[1:]
    if method == COMP_ZLIB:
        return zlib.decompress(payload)
    if method == COMP_BZIP2:
        return bz2.decompress(payload)
    raise NotImplementedError(f'MPQ compression 0x{method:02x}')


class MPQArchive:
    def __init__(self, path_or_data):
        if isinstance(path_or_data, (bytes, bytearray)):
            self._data = bytes(path_or_data)
        else:
            with open(path_or_data, 'rb') as f:
                self._data = f.read()
        self.user_data = None
        offset = 0
        if self._data[:4] == MPQ_USER_DATA_MAGIC:
            user_data_size, header_offset, ud_header_size = \
                struct.unpack_from('<III', self._data, 4)
            self.user_data = self._data[16:16 + ud_header_size]
            offset = header_offset
        if self._data[offset:offset + 4] != MPQ_HEADER_MAGIC:
            raise ValueError('not an MPQ archive')
        (self._hdr_size, self._archive_size, self._fmt, self._block_shift,
         ht_off, bt_off, ht_entries, bt_entries) = \
            struct.unpack_from('<IIHHIIII', self._data, offset + 4)
        self._base = offset
        self._sector_size = 512 << self._block_shift
        self._hash_table = self._read_table(ht_off, ht_entries, '(hash table)')
        self._block_table = self._read_table(bt_off, bt_entries, '(block table)')

    def _read_table(self, off, entries, name):
        raw = self._data[self._base + off:self._base + off + entries * 16]
        dec = decrypt(raw, mpq_hash(name, HASH_FILE_KEY))
        return [struct.unpack_from('<4I', dec, i * 16) for i in range(entries)]

    def _find_hash_entry(self, filename):
        name_a = mpq_hash(filename, HASH_NAME_A)
        name_b = mpq_hash(filename, HASH_NAME_B)
        n = len(self._hash_table)
        start = mpq_hash(filename, HASH_TABLE_OFFSET) & (n - 1)
        for i in range(n):
            entry = self._hash_table[(start + i) % n]
            if entry[0] == name_a and entry[1] == name_b:
                return entry
            if entry[3] == 0xFFFFFFFF:   # empty, never used: stop probing
                return None
        return None

    def read_file(self, filename: str) -> bytes:
        entry = self._find_hash_entry(filename)
        if entry is None:
            raise KeyError(filename)
        block_index = entry[3]
        offset, archived_size, file_size, flags = self._block_table[block_index]
        if not flags & FLAG_EXISTS:
            raise KeyError(filename)
        if flags & FLAG_ENCRYPTED:
            raise NotImplementedError('encrypted MPQ file data')
        if flags & FLAG_IMPLODE:
            raise NotImplementedError('PKWARE-imploded MPQ file data')
        data = self._data[self._base + offset:
                          self._base + offset + archived_size]
        if not flags & FLAG_COMPRESS:
            return data[:file_size]
        if flags & FLAG_SINGLE_UNIT:
            return _decompress(data) if archived_size < file_size \
                else data[:file_size]
        # sectored: u32 offset table (relative to block start), one entry
        # per sector + terminator
        num_sectors = (file_size + self._sector_size - 1) // self._sector_size
        offsets = struct.unpack_from(f'<{num_sectors + 1}I', data, 0)
        out = bytearray()
        remaining = file_size
        for i in range(num_sectors):
            sector = data[offsets[i]:offsets[i + 1]]
            want = min(self._sector_size, remaining)
            if len(sector) < want:         # compressed sector
                sector = _decompress(sector)
            out += sector[:want]
            remaining -= want
        return bytes(out)
