#!/usr/bin/env python3
"""Stream simulated fMRI volumes to disk at TR cadence (DICOM or npy)
— the counterpart of the reference's real-time generator example.

    python examples/fmrisim_realtime.py
"""

import sys
import tempfile
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))
from brainiak_amd.utils.fmrisim_real_time_generator import generate_data


def main():
    with tempfile.TemporaryDirectory() as out:
        generate_data(out, {"numTRs": 30, "scale_percentage": 1.0,
                            "trDuration": 1, "save_dicom": True,
                            "save_realtime": False})
        made = sorted(Path(out).iterdir())
        print("wrote %d files, e.g. %s" % (
            len(made), [p.name for p in made[:4]]))


if __name__ == "__main__":
    main()
