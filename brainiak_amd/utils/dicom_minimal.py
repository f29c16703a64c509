"""Minimal self-contained DICOM Part-10 writer/reader.

The reference's real-time generator writes per-TR DICOM files through
pydicom (ref src/brainiak/utils/fmrisim_real_time_generator.py:187-246);
pydicom is not available in this stack, so this module implements the
small slice of the standard the generator needs: multi-frame secondary
-capture MR images in Explicit-VR Little-Endian transfer syntax, plus a
matching reader for round-trip tests and for real-time pipelines that
consume the stream.

Scope (deliberate): uncompressed uint16 monochrome pixel data only, no
sequences, no character-set handling beyond ASCII.  Files produced here
load in standard tools (the tag layout follows PS3.10 §7.1 and PS3.5
§7.1.2).
"""

import os
import struct
import time

import numpy as np

__all__ = ["write_dicom", "read_dicom"]

_EXPLICIT_VR_LE = "1.2.840.10008.1.2.1"
_SECONDARY_CAPTURE = "1.2.840.10008.5.1.4.1.1.7"
# private-use UID root (not registered; fine for synthetic data)
_UID_ROOT = "1.2.826.0.1.3680043.10.1405"

# VRs whose explicit encoding carries a 2-byte reserved field and a
# 4-byte length (PS3.5 table 7.1-1)
_LONG_VRS = {b"OB", b"OW", b"OF", b"SQ", b"UT", b"UN"}


def _new_uid(suffix=None):
    if suffix is None:
        suffix = f"{int(time.time() * 1e3) % 10 ** 12}.{os.getpid() % 9973}"
    return f"{_UID_ROOT}.{suffix}"


def _pad(value, pad_byte):
    return value + pad_byte if len(value) % 2 else value


def _encode(tag_group, tag_elem, vr, value):
    """One explicit-VR little-endian data element."""
    if isinstance(value, str):
        value = value.encode("ascii")
    if vr == b"UI":
        value = _pad(value, b"\x00")
    elif vr in (b"CS", b"IS", b"DS", b"SH", b"LO", b"DA", b"TM", b"PN"):
        value = _pad(value, b" ")
    head = struct.pack("<HH", tag_group, tag_elem) + vr
    if vr in _LONG_VRS:
        return head + b"\x00\x00" + struct.pack("<I", len(value)) + value
    return head + struct.pack("<H", len(value)) + value


def _us(v):
    return struct.pack("<H", int(v))


def write_dicom(path, volume, instance_number=1, series_number=1,
                tr_seconds=None, description="brainiak_amd fmrisim"):
    """Write ``volume`` as a multi-frame secondary-capture MR DICOM.

    volume : 2-D [rows, cols] or 3-D [rows, cols, frames] array; values
        are clipped to uint16 (the generator's volumes are positive).
    """
    vol = np.asarray(volume)
    if vol.ndim == 2:
        vol = vol[:, :, None]
    if vol.ndim != 3:
        raise ValueError("volume must be 2-D or 3-D")
    rows, cols, frames = vol.shape
    # frame-major pixel layout: frame f = vol[:, :, f], row-major
    pix = np.ascontiguousarray(
        np.transpose(vol, (2, 0, 1))).clip(0, 65535).astype("<u2")
    pixel_bytes = pix.tobytes()
    if len(pixel_bytes) % 2:
        pixel_bytes += b"\x00"

    sop_uid = _new_uid(f"{series_number}.{instance_number}")

    meta = b"".join([
        _encode(0x0002, 0x0001, b"OB", b"\x00\x01"),
        _encode(0x0002, 0x0002, b"UI", _SECONDARY_CAPTURE),
        _encode(0x0002, 0x0003, b"UI", sop_uid),
        _encode(0x0002, 0x0010, b"UI", _EXPLICIT_VR_LE),
        _encode(0x0002, 0x0012, b"UI", _UID_ROOT + ".0.1"),
    ])
    meta = _encode(0x0002, 0x0000, b"UL",
                   struct.pack("<I", len(meta))) + meta

    elems = [
        _encode(0x0008, 0x0016, b"UI", _SECONDARY_CAPTURE),
        _encode(0x0008, 0x0018, b"UI", sop_uid),
        _encode(0x0008, 0x0060, b"CS", "MR"),
        _encode(0x0008, 0x103E, b"LO", description),
        _encode(0x0020, 0x0011, b"IS", str(series_number)),
        _encode(0x0020, 0x0013, b"IS", str(instance_number)),
        _encode(0x0028, 0x0002, b"US", _us(1)),
        _encode(0x0028, 0x0004, b"CS", "MONOCHROME2"),
        _encode(0x0028, 0x0008, b"IS", str(frames)),
        _encode(0x0028, 0x0010, b"US", _us(rows)),
        _encode(0x0028, 0x0011, b"US", _us(cols)),
        _encode(0x0028, 0x0100, b"US", _us(16)),
        _encode(0x0028, 0x0101, b"US", _us(16)),
        _encode(0x0028, 0x0102, b"US", _us(15)),
        _encode(0x0028, 0x0103, b"US", _us(0)),
    ]
    if tr_seconds is not None:
        # RepetitionTime is in milliseconds (DS)
        elems.insert(4, _encode(0x0018, 0x0080, b"DS",
                                f"{tr_seconds * 1000:.1f}"))
    elems.append(_encode(0x7FE0, 0x0010, b"OW", pixel_bytes))

    with open(path, "wb") as f:
        f.write(b"\x00" * 128)
        f.write(b"DICM")
        f.write(meta)
        f.write(b"".join(elems))


def read_dicom(path):
    """Parse a file written by :func:`write_dicom` (or any uncompressed
    explicit-VR-LE single-image file with the same tag subset).

    Returns (volume [rows, cols, frames] uint16, tags dict keyed by
    (group, elem) with decoded scalar/string values).
    """
    data = open(path, "rb").read()
    if data[128:132] != b"DICM":
        raise ValueError("not a DICOM Part-10 file")
    pos = 132
    tags = {}
    pixels = None
    while pos + 8 <= len(data):
        group, elem = struct.unpack_from("<HH", data, pos)
        vr = data[pos + 4:pos + 6]
        if vr in _LONG_VRS:
            length = struct.unpack_from("<I", data, pos + 8)[0]
            vpos = pos + 12
        else:
            length = struct.unpack_from("<H", data, pos + 6)[0]
            vpos = pos + 8
        value = data[vpos:vpos + length]
        pos = vpos + length
        key = (group, elem)
        if key == (0x7FE0, 0x0010):
            pixels = value
        elif vr == b"US":
            tags[key] = struct.unpack("<H", value)[0]
        elif vr in (b"IS", b"DS"):
            txt = value.decode("ascii").strip()
            tags[key] = float(txt) if vr == b"DS" else int(txt)
        elif vr in (b"UI", b"CS", b"LO", b"SH", b"PN"):
            tags[key] = value.decode("ascii").strip("\x00 ")
        else:
            tags[key] = value
    if pixels is None:
        raise ValueError("no PixelData element")
    rows = tags[(0x0028, 0x0010)]
    cols = tags[(0x0028, 0x0011)]
    frames = int(tags.get((0x0028, 0x0008), 1))
    vol = np.frombuffer(pixels[:rows * cols * frames * 2],
                        dtype="<u2").reshape(frames, rows, cols)
    return np.transpose(vol, (1, 2, 0)), tags
