"""Event visualization: count-map rendering to RGB
(parity: ESR:myutils/vis_events/matplotlib_plot_events.py:125-249).

Implemented as direct numpy compositing (red = positive counts, blue =
negative) rather than a matplotlib figure round-trip; PNG saving uses
matplotlib.image when available.
"""

from __future__ import annotations

import numpy as np

__all__ = ["EventVisualizer"]


class EventVisualizer:
    def plot_event_cnt(self, cnt_hwc: np.ndarray, is_save: bool = False,
                       path: str | None = None,
                       is_black_background: bool = True) -> np.ndarray:
        """cnt_hwc: [H, W, 2] (pos, neg) counts -> [H, W, 3] uint8 render."""
        pos = np.asarray(cnt_hwc[..., 0], dtype=np.float64)
        neg = np.asarray(cnt_hwc[..., 1], dtype=np.float64)

        def norm(c):
            m = c.max()
            return c / m if m > 0 else c

        pos_n, neg_n = norm(pos), norm(neg)
        H, W = pos.shape
        if is_black_background:
            img = np.zeros((H, W, 3), dtype=np.float64)
            img[..., 0] = pos_n
            img[..., 2] = neg_n
        else:
            img = np.ones((H, W, 3), dtype=np.float64)
            img[..., 1] -= pos_n + neg_n          # remove green where events
            img[..., 2] -= pos_n                  # pos -> red
            img[..., 0] -= neg_n                  # neg -> blue
        img = (img.clip(0, 1) * 255).astype(np.uint8)
        if is_save and path:
            self._save(img, path)
        return img

    def plot_frame(self, frame: np.ndarray, is_save: bool = False,
                   path: str | None = None) -> np.ndarray:
        img = np.asarray(frame, dtype=np.uint8)
        if img.ndim == 2:
            img = np.stack([img] * 3, axis=-1)
        if is_save and path:
            self._save(img, path)
        return img

    @staticmethod
    def _save(img: np.ndarray, path: str):
        try:
            import matplotlib.image as mpimg
            mpimg.imsave(path, img)
        except Exception:
            # fall back to raw npy dump if matplotlib is unavailable
            np.save(path + ".npy", img)


def plot_event_stack(stack_bhw, is_save=False, path=None):
    """Render a signed event stack [TB, H, W] as a grid of count images
    (parity in spirit with ESR:myutils/vis_events/matplotlib_plot_events.py
    stack plots)."""
    import numpy as np
    stack = np.asarray(stack_bhw, dtype=np.float64)
    TB, H, W = stack.shape
    viz = EventVisualizer()
    tiles = []
    for b in range(TB):
        pos = np.clip(stack[b], 0, None)
        neg = np.clip(-stack[b], 0, None)
        tiles.append(viz.plot_event_cnt(np.stack([pos, neg], axis=-1)))
    cols = int(np.ceil(np.sqrt(TB)))
    rows = int(np.ceil(TB / cols))
    canvas = np.zeros((rows * H, cols * W, 3), dtype=np.uint8)
    for i, t in enumerate(tiles):
        r, c = divmod(i, cols)
        canvas[r * H:(r + 1) * H, c * W:(c + 1) * W] = t
    if is_save and path:
        EventVisualizer._save(canvas, path)
    return canvas
