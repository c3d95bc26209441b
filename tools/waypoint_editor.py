"""Interactive waypoint editor for robot trajectories.

Capability parity with the reference's
``floorplans/spline_paths/point_selector.py`` (PolygonInteractor GUI):
click waypoints on a floorplan, drag to adjust, save as an ``.npy``
array in the normalized [-1, 1] convention TrajectoryLidarDataset
expects. Headless environments can use --demo to write a synthetic
path without the GUI.

Usage:
    python tools/waypoint_editor.py --img floor.png --out path0.npy
    python tools/waypoint_editor.py --demo --out path0.npy
"""

from __future__ import annotations

import argparse
import sys

import numpy as np


class WaypointEditor:
    """Matplotlib click-to-place/drag waypoint editor."""

    def __init__(self, img, out_path):
        import matplotlib.pyplot as plt

        self.img = img
        self.out_path = out_path
        self.points = []
        self.drag_idx = None
        self.fig, self.ax = plt.subplots(figsize=(7, 7))
        self.ax.imshow(img, cmap="gray_r", extent=[-1, 1, -1, 1],
                       origin="lower")
        self.ax.set_title(
            "left-click: add point | drag: move | right-click: delete\n"
            "key s: save (closed loop) | q: quit"
        )
        (self.line,) = self.ax.plot([], [], "o-r", ms=6)
        self.fig.canvas.mpl_connect("button_press_event", self._press)
        self.fig.canvas.mpl_connect("button_release_event",
                                    self._release)
        self.fig.canvas.mpl_connect("motion_notify_event", self._move)
        self.fig.canvas.mpl_connect("key_press_event", self._key)
        plt.show()

    def _redraw(self):
        if self.points:
            arr = np.asarray(self.points)
            self.line.set_data(arr[:, 0], arr[:, 1])
        else:
            self.line.set_data([], [])
        self.fig.canvas.draw_idle()

    def _nearest(self, x, y, tol=0.05):
        if not self.points:
            return None
        arr = np.asarray(self.points)
        d = np.hypot(arr[:, 0] - x, arr[:, 1] - y)
        i = int(np.argmin(d))
        return i if d[i] < tol else None

    def _press(self, ev):
        if ev.inaxes != self.ax:
            return
        idx = self._nearest(ev.xdata, ev.ydata)
        if ev.button == 3 and idx is not None:
            self.points.pop(idx)
        elif ev.button == 1:
            if idx is None:
                self.points.append([ev.xdata, ev.ydata])
            else:
                self.drag_idx = idx
        self._redraw()

    def _release(self, _ev):
        self.drag_idx = None

    def _move(self, ev):
        if self.drag_idx is None or ev.inaxes != self.ax:
            return
        self.points[self.drag_idx] = [ev.xdata, ev.ydata]
        self._redraw()

    def _key(self, ev):
        if ev.key == "s" and len(self.points) >= 3:
            wp = np.asarray(self.points)
            wp = np.vstack([wp, wp[:1]])  # close the loop
            np.save(self.out_path, wp)
            print(f"saved {len(wp)} waypoints -> {self.out_path}")
        elif ev.key == "q":
            import matplotlib.pyplot as plt

            plt.close(self.fig)


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--img", default=None, help="floorplan PNG")
    p.add_argument("--out", required=True)
    p.add_argument("--demo", action="store_true",
                   help="write a synthetic path (no GUI)")
    args = p.parse_args(argv)

    sys.path.insert(0, ".")
    from nn_distributed_training_amd.data.floorplan import (
        synthetic_floorplan,
        synthetic_waypoints,
    )

    if args.demo:
        img = synthetic_floorplan()
        wp = synthetic_waypoints(img, 1)[0]
        np.save(args.out, wp)
        print(f"saved demo waypoints -> {args.out}")
        return

    if args.img:
        from PIL import Image

        img = np.asarray(Image.open(args.img)).astype(float) / 255.0
    else:
        img = synthetic_floorplan()
    WaypointEditor(img, args.out)


if __name__ == "__main__":
    main()
