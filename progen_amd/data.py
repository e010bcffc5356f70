"""Data pipeline: GZIP TFRecord shards + byte-level tokenizer.

Reimplements the reference's TensorFlow-based pipeline
(reference: progen_transformer/data.py) WITHOUT TensorFlow:

  - TFRecord wire format (length / masked-crc32c framing) and the
    tf.train.Example protobuf for the single ``'seq'`` bytes feature are
    hand-encoded/decoded (the schema is fixed: data.py:9-15,25-28);
  - GZIP compression = the whole record stream gzipped, matching
    tf.io.TFRecordOptions(compression_type='GZIP') (data.py:17-21);
  - shard naming and the filename seq-count contract
    ``{idx}.{count}.{type}.tfrecord.gz`` are preserved
    (reference: data.py:46, generate_data.py:142);
  - the collate path (truncate to seq_len, +1 offset, zero right-pad,
    BOS=0 column prepended -> (b, seq_len+1)) matches data.py:30-35,64-70;
  - byte tokenizer: encode = ord(c)+1, decode = chr(t-1); token 0 is
    PAD / BOS / EOS simultaneously (data.py:76-88).
"""

from __future__ import annotations

import gzip
import struct
from contextlib import contextmanager
from functools import partial
from pathlib import Path
from typing import Iterator, List, Optional, Tuple

import numpy as np

# ---------------------------------------------------------------------------
# CRC-32C (Castagnoli) — TFRecord framing checksum
# ---------------------------------------------------------------------------

_CRC_TABLE: Optional[np.ndarray] = None


def _crc32c_table() -> np.ndarray:
    global _CRC_TABLE
    if _CRC_TABLE is None:
        poly = 0x82F63B78  # reflected CRC-32C polynomial
        table = np.zeros(256, dtype=np.uint32)
        for i in range(256):
            crc = i
            for _ in range(8):
                crc = (crc >> 1) ^ (poly if crc & 1 else 0)
            table[i] = crc
        _CRC_TABLE = table
    return _CRC_TABLE


_NATIVE_CRC = None
_NATIVE_CRC_TRIED = False


def _native_crc():
    # the _C extension's slicing-by-8 CRC is ~1000x the Python loop;
    # soft import: data prep must still work before the extension is
    # built (the GPU-op dispatch in ops/dispatch.py stays strict)
    global _NATIVE_CRC, _NATIVE_CRC_TRIED
    if not _NATIVE_CRC_TRIED:
        _NATIVE_CRC_TRIED = True
        try:
            from progen_amd import _C  # noqa: WPS433
            _NATIVE_CRC = _C.crc32c
        except Exception:  # noqa: BLE001
            _NATIVE_CRC = None
    return _NATIVE_CRC


def crc32c(data: bytes) -> int:
    native = _native_crc()
    if native is not None:
        return int(native(data))
    table = _crc32c_table()
    crc = np.uint32(0xFFFFFFFF)
    buf = np.frombuffer(data, dtype=np.uint8)
    tbl = table
    c = int(crc)
    for b in buf.tobytes():  # byte loop; fallback when _C is not built
        c = tbl[(c ^ b) & 0xFF] ^ (c >> 8)
        c = int(c)
    return c ^ 0xFFFFFFFF


def _masked_crc(data: bytes) -> int:
    crc = crc32c(data)
    return ((((crc >> 15) | (crc << 17)) & 0xFFFFFFFF) + 0xA282EAD8) & 0xFFFFFFFF


# ---------------------------------------------------------------------------
# minimal protobuf for tf.train.Example{features{feature{'seq': bytes_list}}}
# ---------------------------------------------------------------------------

def _varint(n: int) -> bytes:
    out = bytearray()
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _read_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not b & 0x80:
            return result, pos
        shift += 7


def _len_delimited(field_no: int, payload: bytes) -> bytes:
    return _varint((field_no << 3) | 2) + _varint(len(payload)) + payload


def encode_example(seq_bytes: bytes) -> bytes:
    """Serialize tf.train.Example with one 'seq' bytes feature
    (wire-compatible with reference data.py:9-15)."""
    bytes_list = _len_delimited(1, seq_bytes)          # BytesList.value
    feature = _len_delimited(1, bytes_list)            # Feature.bytes_list
    map_entry = _len_delimited(1, b"seq") + _len_delimited(2, feature)
    features = _len_delimited(1, map_entry)            # Features.feature
    return _len_delimited(1, features)                 # Example.features


def _skip_field(buf: bytes, pos: int, wire_type: int) -> int:
    if wire_type == 0:
        _, pos = _read_varint(buf, pos)
    elif wire_type == 1:
        pos += 8
    elif wire_type == 2:
        ln, pos = _read_varint(buf, pos)
        pos += ln
    elif wire_type == 5:
        pos += 4
    else:
        raise ValueError(f"unsupported wire type {wire_type}")
    return pos


def _fields(buf: bytes) -> Iterator[Tuple[int, int, bytes]]:
    pos = 0
    while pos < len(buf):
        tag, pos = _read_varint(buf, pos)
        field_no, wire_type = tag >> 3, tag & 7
        if wire_type == 2:
            ln, pos = _read_varint(buf, pos)
            yield field_no, wire_type, buf[pos:pos + ln]
            pos += ln
        else:
            start = pos
            pos = _skip_field(buf, pos, wire_type)
            yield field_no, wire_type, buf[start:pos]


def decode_example(example_bytes: bytes, key: str = "seq") -> bytes:
    """Extract the named bytes feature from a serialized Example
    (the fixed-schema analog of tf.io.parse_single_example,
    reference data.py:25-28)."""
    for fno, _, features in _fields(example_bytes):
        if fno != 1:
            continue
        for fno2, _, entry in _fields(features):
            if fno2 != 1:
                continue
            k: Optional[bytes] = None
            feat: Optional[bytes] = None
            for fno3, _, payload in _fields(entry):
                if fno3 == 1:
                    k = payload
                elif fno3 == 2:
                    feat = payload
            if k == key.encode() and feat is not None:
                for fno4, _, blist in _fields(feat):
                    if fno4 == 1:  # bytes_list
                        for fno5, _, value in _fields(blist):
                            if fno5 == 1:
                                return value
    raise KeyError(f"feature {key!r} not found in Example")


# ---------------------------------------------------------------------------
# TFRecord framing (GZIP stream)
# ---------------------------------------------------------------------------

def write_record(fh, payload: bytes) -> None:
    length = struct.pack("<Q", len(payload))
    fh.write(length)
    fh.write(struct.pack("<I", _masked_crc(length)))
    fh.write(payload)
    fh.write(struct.pack("<I", _masked_crc(payload)))


def read_records(fh) -> Iterator[bytes]:
    while True:
        header = fh.read(8)
        if len(header) < 8:
            return
        (length,) = struct.unpack("<Q", header)
        (lcrc,) = struct.unpack("<I", fh.read(4))
        if lcrc != _masked_crc(header):
            raise IOError("TFRecord length CRC mismatch")
        payload = fh.read(length)
        (pcrc,) = struct.unpack("<I", fh.read(4))
        if pcrc != _masked_crc(payload):
            raise IOError("TFRecord payload CRC mismatch")
        yield payload


def write(writer_fh, values: bytes) -> None:
    """Write one sequence (bytes) as an Example record
    (reference: data.py:9-15)."""
    write_record(writer_fh, encode_example(values))


@contextmanager
def with_tfrecord_writer(path: str):
    """GZIP TFRecord writer context (reference: data.py:17-21)."""
    with gzip.open(path, "wb") as fh:
        yield partial(write, fh)


def iter_tfrecord_file(path: str) -> Iterator[bytes]:
    with gzip.open(path, "rb") as fh:
        for payload in read_records(fh):
            yield decode_example(payload)


# ---------------------------------------------------------------------------
# dataset iterator (reference: data.py:25-72)
# ---------------------------------------------------------------------------

def collate_fn(batch: List[bytes], pad_length: int, offset: int = 0) -> np.ndarray:
    """uint8 bytes -> uint16, truncate, +offset, zero right-pad
    (reference: data.py:30-35)."""
    tensors = [np.frombuffer(el, dtype=np.uint8).astype(np.uint16)[:pad_length] + offset
               for el in batch]
    padded = [np.pad(t, (0, pad_length - t.shape[-1])) for t in tensors]
    return np.stack(padded)


def prefetch_iter(it: Iterator, depth: int = 2) -> Iterator:
    """Background-thread prefetch (the reference pipeline's
    tf.data .prefetch(AUTOTUNE), data.py:62): decompression/collate of
    the next batches overlaps the training step. Exceptions in the
    producer re-raise in the consumer."""
    import queue
    import threading

    q: "queue.Queue" = queue.Queue(maxsize=max(1, depth))
    sentinel = object()

    def worker():
        try:
            for item in it:
                q.put(item)
        except BaseException as e:  # noqa: BLE001 — forwarded to consumer
            q.put(e)
            return
        q.put(sentinel)

    threading.Thread(target=worker, daemon=True).start()
    while True:
        item = q.get()
        if item is sentinel:
            return
        if isinstance(item, BaseException):
            raise item
        yield item


def iterator_from_tfrecords_folder(folder: str, data_type: str = "train"):
    """Returns (num_seqs, iter_fn) (reference: data.py:37-72).

    num_seqs is parsed from the shard FILENAME — field -4 of the
    '.'-split name, i.e. the {count} of '{idx}.{count}.{type}.tfrecord.gz'
    (reference: data.py:46). gs:// paths require google-cloud-storage
    (not available offline) and raise."""
    if folder.startswith("gs://"):
        raise NotImplementedError(
            "gs:// TFRecord folders need google-cloud-storage, which is not "
            "installed in this offline environment")
    folder_p = Path(folder)
    filenames = sorted(str(p) for p in folder_p.glob(f"**/*.{data_type}.tfrecord.gz"))
    num_seqs = sum(int(f.split(".")[-4]) for f in filenames)

    def iter_fn(seq_len: int, batch_size: int, skip: int = 0,
                loop: bool = False, prefetch: int = 2):
        def gen_sequences():
            while True:
                for fname in filenames:
                    yield from iter_tfrecord_file(fname)
                if not loop:
                    return

        gen = gen_sequences()
        for _ in range(skip):
            try:
                next(gen)
            except StopIteration:
                return

        def gen_batches():
            batch: List[bytes] = []
            for seq in gen:
                batch.append(seq)
                if len(batch) == batch_size:
                    yield _finalize_batch(batch, seq_len)
                    batch = []
            if batch:
                yield _finalize_batch(batch, seq_len)

        if prefetch > 0:
            yield from prefetch_iter(gen_batches(), depth=prefetch)
        else:
            yield from gen_batches()

    return num_seqs, iter_fn


def _finalize_batch(batch: List[bytes], seq_len: int) -> np.ndarray:
    seq = collate_fn(batch, pad_length=seq_len, offset=1)
    bos = np.zeros((seq.shape[0], 1), dtype=np.uint16)
    return np.concatenate((bos, seq), axis=1)  # (b, seq_len+1)


# ---------------------------------------------------------------------------
# tokenization (reference: data.py:76-88)
# ---------------------------------------------------------------------------

def encode_token(token: str) -> int:
    return ord(token) + 1


def decode_token(token: int) -> str:
    if token < 0:
        return ""
    return chr(token)


def encode_tokens(tokens: str) -> List[int]:
    return [encode_token(t) for t in tokens]


def decode_tokens(tokens, offset: int = 1) -> str:
    arr = np.asarray(tokens).astype(np.int64) - offset
    return "".join(decode_token(int(t)) for t in arr)
