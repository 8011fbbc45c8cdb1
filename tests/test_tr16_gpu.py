"""Pin ds_read_b64_tr_b16 semantics on silicon (the wgrad staging relies on
this exact mapping): with lane addresses base + lane*8, a 16-lane group
collectively covers a 128-byte block and lane c receives column c of the
block viewed as a 4x16 row-major ushort matrix."""
import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def test_tr16_group_transpose():
    from p2pvg_amd.ops import _hip_ext_loader

    ext = _hip_ext_loader.load()
    out = ext.tr16_probe(0).cpu().numpy().astype(np.uint16)  # addr = lane*8
    for l in range(64):
        group_base = (l // 16) * 64  # elements
        col = l % 16
        expected = [group_base + col + j * 16 for j in range(4)]
        assert list(out[l]) == expected, (l, list(out[l]), expected)


def test_tr16_plain_control():
    from p2pvg_amd.ops import _hip_ext_loader

    ext = _hip_ext_loader.load()
    out = ext.tr16_probe(4).cpu().numpy().astype(np.uint16)  # plain ds_read_b64
    for l in range(64):
        assert list(out[l]) == [l * 4 + j for j in range(4)]
