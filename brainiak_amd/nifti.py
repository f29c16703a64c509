"""Self-contained NIfTI-1 / NIfTI-2 reader/writer.

The reference uses nibabel for all volume IO (ref src/brainiak/io.py:39-168);
nibabel is not part of this stack, so this module implements the subset
the toolkit needs directly on numpy: .nii / .nii.gz, NIfTI-1 and
NIfTI-2 single-file images in either byte order, scalar datatypes,
scl_slope/scl_inter scaling, and sform/qform affines.  ``save`` writes
NIfTI-1 by default or NIfTI-2 with ``version=2``.

``NiftiImage`` intentionally mirrors the parts of nibabel's
``SpatialImage`` API the rest of the package touches: ``get_fdata()``,
``get_data()``, ``affine``, ``shape``, ``header``.
"""

import gzip
import struct
from pathlib import Path
from typing import Union

import numpy as np

__all__ = ["NiftiImage", "load", "save"]

_DTYPES = {
    2: np.uint8,
    4: np.int16,
    8: np.int32,
    16: np.float32,
    64: np.float64,
    256: np.int8,
    512: np.uint16,
    768: np.uint32,
    1024: np.int64,
}
_DTYPE_CODES = {np.dtype(v): k for k, v in _DTYPES.items()}


class NiftiHeader(dict):
    """Dict-backed NIfTI-1 header with a nibabel-ish get_zooms()."""

    def get_zooms(self):
        ndim = int(self["dim"][0])
        return tuple(float(z) for z in self["pixdim"][1:1 + ndim])


class NiftiImage:
    """An in-memory NIfTI volume: data + affine + header."""

    def __init__(self, dataobj, affine=None, header=None):
        self.dataobj = np.asarray(dataobj)
        if affine is None:
            affine = np.eye(4)
        self.affine = np.asarray(affine, dtype=np.float64)
        self.header = header if header is not None else \
            _default_header(self.dataobj)

    @property
    def shape(self):
        return self.dataobj.shape

    def get_fdata(self, dtype=np.float64):
        return np.asarray(self.dataobj, dtype=dtype)

    def get_data(self):  # nibabel-compat alias
        return self.dataobj

    def to_filename(self, path):
        save(self, path)


def _default_header(data):
    hdr = NiftiHeader()
    dim = np.ones(8, dtype=np.int16)
    dim[0] = data.ndim
    dim[1:1 + data.ndim] = data.shape
    hdr["dim"] = dim
    hdr["pixdim"] = np.ones(8, dtype=np.float32)
    hdr["datatype"] = _DTYPE_CODES.get(np.dtype(data.dtype), 16)
    hdr["scl_slope"] = 1.0
    hdr["scl_inter"] = 0.0
    return hdr


def _open_maybe_gz(path, mode):
    path = str(path)
    if path.endswith(".gz"):
        return gzip.open(path, mode)
    return open(path, mode)


def _parse_n1(raw, en, path):
    """NIfTI-1 header fields (348-byte layout), endianness ``en``."""
    hdr = NiftiHeader()
    hdr["nifti_version"] = 1
    hdr["dim"] = np.frombuffer(raw, en + "i2", count=8, offset=40).copy()
    hdr["datatype"] = struct.unpack_from(en + "h", raw, 70)[0]
    hdr["bitpix"] = struct.unpack_from(en + "h", raw, 72)[0]
    hdr["pixdim"] = np.frombuffer(raw, en + "f4", count=8,
                                  offset=76).copy()
    hdr["vox_offset"] = struct.unpack_from(en + "f", raw, 108)[0]
    hdr["scl_slope"] = struct.unpack_from(en + "f", raw, 112)[0]
    hdr["scl_inter"] = struct.unpack_from(en + "f", raw, 116)[0]
    hdr["qform_code"] = struct.unpack_from(en + "h", raw, 252)[0]
    hdr["sform_code"] = struct.unpack_from(en + "h", raw, 254)[0]
    hdr["quatern"] = struct.unpack_from(en + "3f", raw, 256)
    hdr["qoffset"] = struct.unpack_from(en + "3f", raw, 268)
    hdr["srow"] = np.frombuffer(raw, en + "f4", count=12,
                                offset=280).reshape(3, 4).copy()
    magic = raw[344:348]
    if magic[:3] not in (b"n+1", b"ni1"):
        raise ValueError(f"{path}: bad NIfTI-1 magic {magic!r}")
    hdr["min_offset"] = 352
    return hdr


def _parse_n2(raw, en, path):
    """NIfTI-2 header fields (540-byte layout), endianness ``en``."""
    hdr = NiftiHeader()
    hdr["nifti_version"] = 2
    magic = raw[4:12]
    if magic[:3] not in (b"n+2", b"ni2"):
        raise ValueError(f"{path}: bad NIfTI-2 magic {magic!r}")
    hdr["datatype"] = struct.unpack_from(en + "h", raw, 12)[0]
    hdr["bitpix"] = struct.unpack_from(en + "h", raw, 14)[0]
    hdr["dim"] = np.frombuffer(raw, en + "i8", count=8, offset=16).copy()
    hdr["pixdim"] = np.frombuffer(raw, en + "f8", count=8,
                                  offset=104).copy()
    hdr["vox_offset"] = struct.unpack_from(en + "q", raw, 168)[0]
    hdr["scl_slope"] = struct.unpack_from(en + "d", raw, 176)[0]
    hdr["scl_inter"] = struct.unpack_from(en + "d", raw, 184)[0]
    hdr["qform_code"] = struct.unpack_from(en + "i", raw, 344)[0]
    hdr["sform_code"] = struct.unpack_from(en + "i", raw, 348)[0]
    hdr["quatern"] = struct.unpack_from(en + "3d", raw, 352)
    hdr["qoffset"] = struct.unpack_from(en + "3d", raw, 376)
    hdr["srow"] = np.frombuffer(raw, en + "f8", count=12,
                                offset=400).reshape(3, 4).copy()
    hdr["min_offset"] = 544
    return hdr


def load(path: Union[str, Path]) -> NiftiImage:
    """Load a .nii / .nii.gz file (NIfTI-1 or NIfTI-2, either byte
    order)."""
    with _open_maybe_gz(path, "rb") as f:
        raw = f.read()
    if len(raw) < 348:
        raise ValueError(f"{path}: too short to be a NIfTI file")
    hdr = None
    for en in ("<", ">"):
        size = struct.unpack_from(en + "i", raw, 0)[0]
        if size == 348:
            hdr = _parse_n1(raw, en, path)
            break
        if size == 540:
            hdr = _parse_n2(raw, en, path)
            break
    if hdr is None:
        raise ValueError(f"{path}: not a NIfTI-1/2 file (sizeof_hdr="
                         f"{struct.unpack_from('<i', raw, 0)[0]})")

    ndim = int(hdr["dim"][0])
    shape = tuple(int(d) for d in hdr["dim"][1:1 + ndim])
    dtype = _DTYPES.get(hdr["datatype"])
    if dtype is None:
        raise ValueError(f"{path}: unsupported NIfTI datatype "
                         f"{hdr['datatype']}")
    offset = max(int(hdr["vox_offset"]), hdr["min_offset"])
    count = int(np.prod(shape))
    data = np.frombuffer(raw, np.dtype(dtype).newbyteorder(en),
                         count=count, offset=offset)
    # NIfTI data is Fortran-ordered (x fastest)
    data = data.reshape(shape, order="F")
    data = data.astype(data.dtype.newbyteorder("="))

    slope, inter = float(hdr["scl_slope"]), float(hdr["scl_inter"])
    if slope not in (0.0, 1.0) or inter != 0.0:
        data = data * (slope if slope != 0.0 else 1.0) + inter

    if hdr["sform_code"] > 0:
        affine = np.eye(4)
        affine[:3, :] = hdr["srow"]
    elif hdr["qform_code"] > 0:
        affine = _affine_from_quatern(hdr)
    else:
        affine = np.diag(list(hdr["pixdim"][1:4]) + [1.0])
    return NiftiImage(data, affine, hdr)


def _affine_from_quatern(hdr):
    b, c, d = hdr["quatern"]
    qx, qy, qz = hdr["qoffset"]
    a2 = 1.0 - (b * b + c * c + d * d)
    a = np.sqrt(max(a2, 0.0))
    R = np.array([
        [a * a + b * b - c * c - d * d, 2 * (b * c - a * d),
         2 * (b * d + a * c)],
        [2 * (b * c + a * d), a * a + c * c - b * b - d * d,
         2 * (c * d - a * b)],
        [2 * (b * d - a * c), 2 * (c * d + a * b),
         a * a + d * d - b * b - c * c],
    ])
    pixdim = hdr["pixdim"]
    qfac = -1.0 if pixdim[0] == -1.0 else 1.0
    zooms = np.array([pixdim[1], pixdim[2], pixdim[3] * qfac])
    affine = np.eye(4)
    affine[:3, :3] = R * zooms
    affine[:3, 3] = [qx, qy, qz]
    return affine


def save(img: NiftiImage, path: Union[str, Path],
         version: int = 1) -> None:
    """Write a NiftiImage as NIfTI-1 (default) or NIfTI-2
    (``version=2``); .nii or .nii.gz by extension."""
    data = np.asarray(img.dataobj)
    dtype = np.dtype(data.dtype)
    if dtype not in _DTYPE_CODES:
        data = data.astype(np.float32)
        dtype = np.dtype(np.float32)
    code = _DTYPE_CODES[dtype]
    if version == 2:
        _save_n2(img, data, dtype, code, path)
        return
    if version != 1:
        raise ValueError("version must be 1 or 2")

    hdr = bytearray(348)
    struct.pack_into("<i", hdr, 0, 348)
    dim = np.ones(8, dtype=np.int16)
    dim[0] = data.ndim
    dim[1:1 + data.ndim] = data.shape
    struct.pack_into("<8h", hdr, 40, *dim)
    struct.pack_into("<h", hdr, 70, code)
    struct.pack_into("<h", hdr, 72, dtype.itemsize * 8)
    pixdim = np.ones(8, dtype=np.float32)
    if isinstance(img.header, dict) and "pixdim" in img.header:
        src = np.asarray(img.header["pixdim"], dtype=np.float32)
        pixdim[:len(src)] = src
    struct.pack_into("<8f", hdr, 76, *pixdim)
    struct.pack_into("<f", hdr, 108, 352.0)   # vox_offset
    struct.pack_into("<f", hdr, 112, 1.0)     # scl_slope
    struct.pack_into("<f", hdr, 116, 0.0)     # scl_inter
    struct.pack_into("<h", hdr, 252, 0)       # qform_code
    struct.pack_into("<h", hdr, 254, 2)       # sform_code = aligned
    affine = np.asarray(img.affine, dtype=np.float32)
    struct.pack_into("<12f", hdr, 280, *affine[:3, :].ravel())
    hdr[344:348] = b"n+1\x00"

    payload = bytes(hdr) + b"\x00" * 4 + data.astype(
        dtype.newbyteorder("<")).tobytes(order="F")
    with _open_maybe_gz(path, "wb") as f:
        f.write(payload)


def _save_n2(img, data, dtype, code, path):
    hdr = bytearray(540)
    struct.pack_into("<i", hdr, 0, 540)
    hdr[4:12] = b"n+2\x00\r\n\x1a\n"
    struct.pack_into("<h", hdr, 12, code)
    struct.pack_into("<h", hdr, 14, dtype.itemsize * 8)
    dim = np.ones(8, dtype=np.int64)
    dim[0] = data.ndim
    dim[1:1 + data.ndim] = data.shape
    struct.pack_into("<8q", hdr, 16, *dim)
    pixdim = np.ones(8, dtype=np.float64)
    if isinstance(img.header, dict) and "pixdim" in img.header:
        src = np.asarray(img.header["pixdim"], dtype=np.float64)
        pixdim[:len(src)] = src
    struct.pack_into("<8d", hdr, 104, *pixdim)
    struct.pack_into("<q", hdr, 168, 544)     # vox_offset
    struct.pack_into("<d", hdr, 176, 1.0)     # scl_slope
    struct.pack_into("<d", hdr, 184, 0.0)     # scl_inter
    struct.pack_into("<i", hdr, 344, 0)       # qform_code
    struct.pack_into("<i", hdr, 348, 2)       # sform_code = aligned
    affine = np.asarray(img.affine, dtype=np.float64)
    struct.pack_into("<12d", hdr, 400, *affine[:3, :].ravel())
    payload = bytes(hdr) + b"\x00" * 4 + data.astype(
        dtype.newbyteorder("<")).tobytes(order="F")
    with _open_maybe_gz(path, "wb") as f:
        f.write(payload)
