"""Minimal video IO via the ffmpeg binary (raw RGB pipe).

The reference reads/writes video with cv2.VideoCapture / cv2.VideoWriter
(inference.py:238-256); OpenCV is unavailable offline, so this pipes raw
frames through ffmpeg when the binary exists and raises a clear error
otherwise.
"""

import json
import shutil
import subprocess


def _require_ffmpeg():
    path = shutil.which("ffmpeg")
    if path is None:
        raise RuntimeError(
            "Video IO requires the ffmpeg binary, which is not installed in "
            "this environment (and cv2 is unavailable offline). Extract "
            "frames to a directory of images and run inference on that."
        )
    return path


class FFmpegReader:
    def __init__(self, path):
        _require_ffmpeg()
        probe = shutil.which("ffprobe")
        if probe is None:
            raise RuntimeError("ffprobe not found")
        info = json.loads(
            subprocess.check_output(
                [probe, "-v", "error", "-select_streams", "v:0",
                 "-show_entries", "stream=width,height,r_frame_rate",
                 "-of", "json", str(path)]
            )
        )["streams"][0]
        self.width = int(info["width"])
        self.height = int(info["height"])
        num, den = info["r_frame_rate"].split("/")
        self.fps = float(num) / float(den)
        self._proc = subprocess.Popen(
            ["ffmpeg", "-v", "error", "-i", str(path), "-f", "rawvideo",
             "-pix_fmt", "rgb24", "-"],
            stdout=subprocess.PIPE,
        )

    def __iter__(self):
        import numpy as np

        nbytes = self.width * self.height * 3
        while True:
            buf = self._proc.stdout.read(nbytes)
            if len(buf) < nbytes:
                break
            yield np.frombuffer(buf, dtype=np.uint8).reshape(
                self.height, self.width, 3
            )
        self._proc.stdout.close()
        self._proc.wait()


class FFmpegWriter:
    def __init__(self, path, width, height, fps):
        _require_ffmpeg()
        self._proc = subprocess.Popen(
            ["ffmpeg", "-v", "error", "-y", "-f", "rawvideo", "-pix_fmt",
             "rgb24", "-s", f"{width}x{height}", "-r", str(fps), "-i", "-",
             "-pix_fmt", "yuv420p", str(path)],
            stdin=subprocess.PIPE,
        )

    def write(self, frame):
        self._proc.stdin.write(frame.astype("uint8").tobytes())

    def close(self):
        self._proc.stdin.close()
        self._proc.wait()
