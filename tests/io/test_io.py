"""NIfTI-2 / endianness breadth tests."""

import numpy as np
import pytest


# -- NIfTI-2 / endianness breadth ------------------------

def test_nifti2_roundtrip(tmp_path):
    from brainiak_amd import nifti
    rng = np.random.RandomState(0)
    data = rng.rand(7, 6, 5, 3).astype(np.float32)
    affine = np.diag([2.0, 2.0, 3.0, 1.0])
    affine[:3, 3] = [-10, -20, -5]
    img = nifti.NiftiImage(data, affine)
    p = str(tmp_path / "v2.nii.gz")
    nifti.save(img, p, version=2)
    back = nifti.load(p)
    assert back.header["nifti_version"] == 2
    assert back.shape == data.shape
    assert np.allclose(back.get_fdata(), data, atol=1e-6)
    assert np.allclose(back.affine, affine)


def test_nifti2_int_dtypes(tmp_path):
    from brainiak_amd import nifti
    for dt in (np.int16, np.int32, np.uint8, np.float64):
        data = (np.arange(60).reshape(5, 4, 3) % 120).astype(dt)
        p = str(tmp_path / f"d_{np.dtype(dt).name}.nii")
        nifti.save(nifti.NiftiImage(data), p, version=2)
        back = nifti.load(p)
        assert back.get_data().dtype == dt
        assert np.array_equal(back.get_data(), data)


def test_nifti1_big_endian(tmp_path):
    """Hand-built big-endian NIfTI-1 file loads correctly."""
    import struct

    from brainiak_amd import nifti
    data = np.arange(24, dtype=np.int16).reshape(2, 3, 4)
    hdr = bytearray(348)
    struct.pack_into(">i", hdr, 0, 348)
    dim = np.ones(8, dtype=np.int16)
    dim[0] = 3
    dim[1:4] = data.shape
    struct.pack_into(">8h", hdr, 40, *dim)
    struct.pack_into(">h", hdr, 70, 4)        # int16
    struct.pack_into(">h", hdr, 72, 16)
    struct.pack_into(">8f", hdr, 76, *np.ones(8, dtype=np.float32))
    struct.pack_into(">f", hdr, 108, 352.0)
    struct.pack_into(">f", hdr, 112, 1.0)
    struct.pack_into(">h", hdr, 254, 1)       # sform
    aff = np.eye(4)[:3, :].ravel()
    struct.pack_into(">12f", hdr, 280, *aff)
    hdr[344:348] = b"n+1\x00"
    p = str(tmp_path / "be.nii")
    with open(p, "wb") as f:
        f.write(bytes(hdr) + b"\x00" * 4
                + data.astype(">i2").tobytes(order="F"))
    img = nifti.load(p)
    assert img.shape == (2, 3, 4)
    assert np.array_equal(img.get_data(), data)


def test_nifti_scl_slope_applied(tmp_path):
    import struct

    from brainiak_amd import nifti
    data = np.arange(12, dtype=np.int16).reshape(3, 4)
    img = nifti.NiftiImage(data)
    p = str(tmp_path / "scl.nii")
    nifti.save(img, p)
    # patch scl_slope/inter in place
    raw = bytearray(open(p, "rb").read())
    struct.pack_into("<f", raw, 112, 2.0)
    struct.pack_into("<f", raw, 116, 10.0)
    open(p, "wb").write(raw)
    back = nifti.load(p)
    assert np.allclose(back.get_fdata(), data * 2.0 + 10.0)


def test_nifti_rejects_garbage(tmp_path):
    from brainiak_amd import nifti
    p = str(tmp_path / "x.nii")
    open(p, "wb").write(b"\x00" * 400)
    with pytest.raises(ValueError):
        nifti.load(p)
