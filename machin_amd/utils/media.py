"""Image / video logging helpers.

Parity target: reference ``machin/utils/media.py``: save numpy frames
as images and frame sequences as videos. moviepy is absent in the
ROCm image, so videos are written as animated GIFs via matplotlib, or
as .npz frame archives when matplotlib is unavailable.
"""
import os
from typing import List

import numpy as np


def _check_frame(frame: np.ndarray):
    if frame.ndim == 2:
        return frame
    if frame.ndim == 3 and frame.shape[-1] in (1, 3, 4):
        return frame
    raise ValueError(f"Unsupported frame shape {frame.shape}")


def show_image(image: np.ndarray, show_normalized: bool = True,
               pause_time: float = 0.01, title: str = ""):
    import matplotlib.pyplot as plt

    _check_frame(image)
    plt.figure(title or "image")
    if show_normalized:
        vmin, vmax = float(image.min()), float(image.max())
        plt.imshow(image, vmin=vmin, vmax=vmax)
    else:
        plt.imshow(image)
    plt.pause(pause_time)


def create_image(image: np.ndarray, path: str, filename: str,
                 extension: str = "png"):
    import matplotlib.image as mpimg

    _check_frame(image)
    os.makedirs(path, exist_ok=True)
    img = image.astype(np.float32)
    if img.max() > 1.0:
        img = img / 255.0
    mpimg.imsave(os.path.join(path, f"{filename}.{extension}"), img)


def create_image_subproc(image: np.ndarray, path: str, filename: str,
                         extension: str = "png", daemon: bool = True):
    """Write the image from a child process; returns its wait fn."""
    from ..parallel.process import Process

    p = Process(
        target=create_image, args=(image, path, filename, extension),
        daemon=daemon,
    )
    p.start()

    def wait():
        p.join()
        p.watch()

    return wait


def create_video(
    frames: List[np.ndarray],
    path: str,
    filename: str,
    extension: str = "gif",
    fps: int = 25,
):
    """Write frames as an animated GIF (no moviepy in this image)."""
    if not frames:
        return
    os.makedirs(path, exist_ok=True)
    out_path = os.path.join(path, f"{filename}.{extension}")
    if extension == "npz":
        np.savez_compressed(out_path, frames=np.stack(frames))
        return
    try:
        import matplotlib.animation as animation
        import matplotlib.pyplot as plt

        fig = plt.figure(frameon=False)
        fig.set_size_inches(
            frames[0].shape[1] / 100, frames[0].shape[0] / 100
        )
        ax = plt.Axes(fig, [0.0, 0.0, 1.0, 1.0])
        ax.set_axis_off()
        fig.add_axes(ax)
        images = [[ax.imshow(_check_frame(f), animated=True)]
                  for f in frames]
        ani = animation.ArtistAnimation(
            fig, images, interval=1000 // fps, blit=True
        )
        ani.save(out_path, writer="pillow", fps=fps)
        plt.close(fig)
    except Exception:  # noqa: BLE001 - fall back to raw archive
        np.savez_compressed(
            os.path.join(path, f"{filename}.npz"), frames=np.stack(frames)
        )


def create_video_subproc(
    frames: List[np.ndarray],
    path: str,
    filename: str,
    extension: str = "gif",
    fps: int = 25,
    daemon: bool = True,
):
    from ..parallel.process import Process

    p = Process(
        target=create_video,
        args=(frames, path, filename, extension, fps),
        daemon=daemon,
    )
    p.start()

    def wait():
        p.join()
        p.watch()

    return wait
