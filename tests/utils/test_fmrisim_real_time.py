import os

import numpy as np
import pytest

from brainiak_amd.utils import fmrisim_real_time_generator as rtg


def test_generate_data(tmp_path):
    np.random.seed(0)
    settings = {'numTRs': 12, 'trDuration': 1, 'isi': 2, 'burn_in': 2,
                'event_duration': 3, 'scale_percentage': 1.0,
                'dimensions': (12, 12, 10), 'save_realtime': False}
    rtg.generate_data(str(tmp_path), settings)
    files = sorted(os.listdir(tmp_path))
    assert 'mask.npy' in files and 'labels.npy' in files
    vols = [f for f in files if f.startswith('rt_')]
    assert len(vols) == 12
    v0 = np.load(tmp_path / vols[0])
    assert v0.shape == (12, 12, 10)
    assert v0.dtype == np.int32
    labels = np.load(tmp_path / 'labels.npy')
    assert set(np.unique(labels)) <= {0.0, 1.0, 2.0}
    mask = np.load(tmp_path / 'mask.npy')
    assert mask.sum() > 0


def test_multivariate_and_different_rois(tmp_path):
    np.random.seed(1)
    settings = {'numTRs': 8, 'trDuration': 1, 'isi': 1, 'burn_in': 1,
                'event_duration': 2, 'dimensions': (10, 10, 8),
                'multivariate_pattern': True, 'different_ROIs': True}
    rtg.generate_data(str(tmp_path), settings)
    assert len([f for f in os.listdir(tmp_path)
                if f.startswith('rt_')]) == 8


def test_generate_data_dicom_stream(tmp_path):
    """save_dicom=True writes valid Part-10 files that round-trip
    through the in-package reader."""
    from brainiak_amd.utils.dicom_minimal import read_dicom
    out = str(tmp_path / "rtd")
    settings = {'numTRs': 4, 'trDuration': 1, 'save_dicom': True,
                'save_realtime': False, 'dimensions': (12, 12, 10),
                'different_ROIs': False, 'multivariate_pattern': False,
                'scale_percentage': 1.0}
    rtg.generate_data(out, settings)
    files = sorted(p for p in os.listdir(out) if p.endswith(".dcm"))
    assert len(files) == 4
    vol, tags = read_dicom(os.path.join(out, files[0]))
    assert vol.shape == (12, 12, 10)
    assert tags[(0x0028, 0x0010)] == 12          # Rows
    assert tags[(0x0008, 0x0060)] == "MR"        # Modality
    assert tags[(0x0018, 0x0080)] == 1000.0      # RepetitionTime ms
    npy = np.load(os.path.join(out, "mask.npy"))
    assert npy.shape == (12, 12, 10)


def test_dicom_writer_roundtrip_values(tmp_path):
    from brainiak_amd.utils.dicom_minimal import read_dicom, write_dicom
    vol = (np.random.rand(7, 9, 5) * 1000).astype(np.int32)
    f = str(tmp_path / "x.dcm")
    write_dicom(f, vol, instance_number=3, series_number=2)
    got, tags = read_dicom(f)
    assert np.array_equal(got.astype(np.int32), vol)
    assert tags[(0x0020, 0x0013)] == 3
    # Part-10 magic present
    raw = open(f, "rb").read()
    assert raw[128:132] == b"DICM"
