"""Video frame IO for the optical-flow pipeline.

Thin OpenCV shims: decode a video into RGB frames (and consecutive-frame
pairs, the flow model's input unit), and encode predicted-flow renderings back
to mp4. cv2 is imported lazily so environments without OpenCV can still use
the rest of the vision stack. Behavior mirrors the reference's
data/vision/video_utils.py.
"""
from __future__ import annotations

import os
from typing import List, Tuple

import numpy as np


def _cv2():
    try:
        import cv2
        return cv2
    except ImportError as e:
        raise ImportError("video_utils requires OpenCV (cv2)") from e


def read_video_frames(video_path: str) -> List[np.ndarray]:
    """Decode every frame of ``video_path`` as an RGB ndarray."""
    cv2 = _cv2()
    if not os.path.exists(video_path):
        raise ValueError(f"Invalid video path supplied. Path '{video_path}' does not exist.")
    capture = cv2.VideoCapture(video_path)
    decoded: List[np.ndarray] = []
    try:
        while capture.isOpened():
            ok, bgr = capture.read()
            if not ok:
                break
            decoded.append(cv2.cvtColor(bgr, cv2.COLOR_BGR2RGB))
    finally:
        if capture is not None:
            capture.release()
    return decoded


def read_video_frame_pairs(video_path: str) -> List[Tuple[np.ndarray, np.ndarray]]:
    """Consecutive frame pairs (t, t+1) — one optical-flow model input each."""
    frames = read_video_frames(video_path)
    return [(a, b) for a, b in zip(frames, frames[1:])]


def write_video(video_path: str, frames: List[np.ndarray], fps: int) -> None:
    """Encode RGB ``frames`` to an mp4 at ``fps``."""
    cv2 = _cv2()
    if os.path.splitext(video_path)[1] != ".mp4":
        raise ValueError("Invalid video path supplied. Only files of type 'mp4' are supported.")
    height, width = frames[0].shape[:2]
    sink = cv2.VideoWriter(video_path, cv2.VideoWriter_fourcc(*"mp4v"), fps, (width, height))
    try:
        for rgb in frames:
            sink.write(cv2.cvtColor(rgb, cv2.COLOR_RGB2BGR))
    finally:
        sink.release()
