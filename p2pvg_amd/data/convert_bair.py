"""Offline BAIR converter: TFRecord -> PNG frame directories.

Capability parity with reference data/convert_bair.py (which requires
TensorFlow 1.x); this version parses the TFRecord framing and the
tf.train.Example protobuf wire format directly, so it runs with no TF
dependency. Layout produced matches the reference exactly:
  <data_dir>/processed_data/{train,test}/<record-stem>/<k>/<i>.png

Usage: python -m p2pvg_amd.data.convert_bair --data_dir <root with
softmotion30_44k/{train,test}>
"""
from __future__ import annotations

import argparse
import glob
import os
import struct
from typing import Dict, Iterator, List, Tuple

import numpy as np


def _read_varint(buf: bytes, pos: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[pos]
        pos += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            return result, pos
        shift += 7


def _parse_fields(buf: bytes) -> Iterator[Tuple[int, int, bytes]]:
    """Yield (field_number, wire_type, payload) for length-delimited and
    varint fields of one protobuf message."""
    pos = 0
    n = len(buf)
    while pos < n:
        tag, pos = _read_varint(buf, pos)
        field, wire = tag >> 3, tag & 7
        if wire == 2:  # length-delimited
            ln, pos = _read_varint(buf, pos)
            yield field, wire, buf[pos : pos + ln]
            pos += ln
        elif wire == 0:  # varint
            v, pos = _read_varint(buf, pos)
            yield field, wire, v  # type: ignore[misc]
        elif wire == 5:  # 32-bit
            yield field, wire, buf[pos : pos + 4]
            pos += 4
        elif wire == 1:  # 64-bit
            yield field, wire, buf[pos : pos + 8]
            pos += 8
        else:
            raise ValueError(f"unsupported wire type {wire}")


def parse_example_bytes_features(record: bytes) -> Dict[str, List[bytes]]:
    """Extract {feature_name: [bytes values]} from a tf.train.Example."""
    out: Dict[str, List[bytes]] = {}
    for f, w, payload in _parse_fields(record):
        if f != 1 or w != 2:  # Example.features
            continue
        for f2, w2, entry in _parse_fields(payload):
            if f2 != 1 or w2 != 2:  # Features.feature map entry
                continue
            key = None
            vals: List[bytes] = []
            for f3, w3, kv in _parse_fields(entry):
                if f3 == 1 and w3 == 2:
                    key = kv.decode("utf-8", "replace")
                elif f3 == 2 and w3 == 2:  # Feature
                    for f4, w4, flist in _parse_fields(kv):
                        if f4 == 1 and w4 == 2:  # BytesList
                            for f5, w5, v in _parse_fields(flist):
                                if f5 == 1 and w5 == 2:
                                    vals.append(v)
            if key is not None and vals:
                out[key] = vals
    return out


def iter_tfrecord(path: str) -> Iterator[bytes]:
    """Yield raw records from a TFRecord file (CRCs skipped, not verified)."""
    with open(path, "rb") as f:
        while True:
            hdr = f.read(8)
            if len(hdr) < 8:
                return
            (length,) = struct.unpack("<Q", hdr)
            f.read(4)  # length crc
            payload = f.read(length)
            f.read(4)  # payload crc
            if len(payload) < length:
                return
            yield payload


def convert_split(data_dir: str, dname: str, image_key: str = "image_aux1",
                  n_frames: int = 30, size: int = 64) -> int:
    from PIL import Image

    src = os.path.join(data_dir, "softmotion30_44k", dname)
    files = sorted(glob.glob(os.path.join(src, "*")))
    if not files:
        raise RuntimeError(f"no TFRecord files under {src}")
    n_clips = 0
    for fpath in files:
        stem = os.path.basename(fpath)
        if stem.endswith(".tfrecords"):
            stem = stem[: -len(".tfrecords")]
        k = 0
        for record in iter_tfrecord(fpath):
            feats = parse_example_bytes_features(record)
            k += 1
            out_dir = os.path.join(
                data_dir, "processed_data", dname, stem, str(k)
            )
            os.makedirs(out_dir, exist_ok=True)
            for i in range(n_frames):
                key = f"{i}/{image_key}/encoded"
                if key not in feats:
                    continue
                raw = feats[key][0]
                arr = np.frombuffer(raw, dtype=np.uint8)
                if arr.size == size * size * 3:  # raw RGB bytes
                    img = Image.frombytes("RGB", (size, size), raw)
                else:  # PNG/JPEG-encoded
                    import io

                    img = Image.open(io.BytesIO(raw)).convert("RGB")
                img.save(os.path.join(out_dir, f"{i}.png"))
            n_clips += 1
        print(f"[convert_bair] {dname}: {stem} ({k} clips)")
    return n_clips


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--data_dir", required=True,
                    help="directory containing softmotion30_44k/{train,test}")
    args = ap.parse_args()
    for dname in ("test", "train"):
        n = convert_split(args.data_dir, dname)
        print(f"[convert_bair] {dname}: {n} clips converted")


if __name__ == "__main__":
    main()
