"""BAIR TFRecord converter: round-trip against a hand-encoded Example."""
import struct

import numpy as np

from p2pvg_amd.data.convert_bair import (
    convert_split,
    iter_tfrecord,
    parse_example_bytes_features,
)


def _varint(n: int) -> bytes:
    out = b""
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out += bytes([b | 0x80])
        else:
            return out + bytes([b])


def _ld(field: int, payload: bytes) -> bytes:
    return _varint((field << 3) | 2) + _varint(len(payload)) + payload


def encode_example(features: dict) -> bytes:
    entries = b""
    for key, values in features.items():
        bytes_list = b"".join(_ld(1, v) for v in values)
        feature = _ld(1, bytes_list)
        entry = _ld(1, key.encode()) + _ld(2, feature)
        entries += _ld(1, entry)
    return _ld(1, entries)  # Example.features


def write_tfrecord(path, records):
    with open(path, "wb") as f:
        for r in records:
            f.write(struct.pack("<Q", len(r)))
            f.write(b"\x00" * 4)
            f.write(r)
            f.write(b"\x00" * 4)


def test_parse_roundtrip(tmp_path):
    feats = {
        "0/image_aux1/encoded": [b"\x01\x02\x03" * 4096],
        "1/image_aux1/encoded": [b"\x04\x05\x06" * 4096],
        "meta": [b"hello"],
    }
    rec = encode_example(feats)
    p = tmp_path / "a.tfrecords"
    write_tfrecord(p, [rec, rec])

    records = list(iter_tfrecord(str(p)))
    assert len(records) == 2
    parsed = parse_example_bytes_features(records[0])
    assert parsed["meta"] == [b"hello"]
    assert parsed["0/image_aux1/encoded"][0] == feats["0/image_aux1/encoded"][0]


def test_convert_split_writes_pngs(tmp_path):
    rng = np.random.RandomState(0)
    frames = {
        f"{i}/image_aux1/encoded": [rng.randint(0, 255, 64 * 64 * 3, dtype=np.uint8).tobytes()]
        for i in range(3)
    }
    rec = encode_example(frames)
    src = tmp_path / "softmotion30_44k" / "train"
    src.mkdir(parents=True)
    write_tfrecord(src / "traj_0_to_9.tfrecords", [rec])

    n = convert_split(str(tmp_path), "train", n_frames=3)
    assert n == 1
    out = tmp_path / "processed_data" / "train" / "traj_0_to_9" / "1"
    assert (out / "0.png").exists() and (out / "2.png").exists()

    from PIL import Image

    img = np.asarray(Image.open(out / "0.png"))
    ref = np.frombuffer(frames["0/image_aux1/encoded"][0], dtype=np.uint8).reshape(64, 64, 3)
    assert (img == ref).all()
