#!/usr/bin/env python3
"""ONNX/TorchScript export for the keypoint model (reference:
pose_estimation/Insulator/export.py) — thin wrapper over the shared deploy
exporter."""
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parents[3]

if __name__ == "__main__":
    sys.exit(subprocess.call(
        [sys.executable, str(REPO / "projects/others/deploy/export_onnx.py"),
         "--model", "hrnet_w18_pose", *sys.argv[1:]]))
