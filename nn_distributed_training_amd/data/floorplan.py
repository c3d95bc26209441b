"""Synthetic floorplans and robot trajectories.

The reference ships a floorplan PNG + hand-drawn waypoint files
(floorplans/32_data/). This environment has no data files, so the density
workload runs on procedurally generated floorplans: a bordered box with
random interior walls, plus smooth random loop trajectories through free
space. A PNG path can still be used via data.lidar.Lidar2D when present.
"""

from __future__ import annotations

import numpy as np


def synthetic_floorplan(
    nx: int = 256,
    ny: int = 256,
    num_walls: int = 10,
    border_width: int = 16,
    seed: int = 0,
) -> np.ndarray:
    """Occupancy image in [0,1], shape [ny, nx]; 1 = wall.

    Square images only: the vertical-wall branch reuses the x-sampled
    bounds as row indices (row/col symmetric when nx == ny, and all
    shipped configs expose a single ``floorplan_size``). Kept as-is so
    seeded maps stay identical to the committed example/benchmark
    artifacts.
    """
    if nx != ny:
        raise ValueError("synthetic_floorplan requires nx == ny")
    rng = np.random.default_rng(seed)
    img = np.zeros((ny, nx), dtype=float)
    img[:border_width, :] = 1.0
    img[-border_width:, :] = 1.0
    img[:, :border_width] = 1.0
    img[:, -border_width:] = 1.0
    thick = max(2, nx // 64)
    for _ in range(num_walls):
        horizontal = rng.random() < 0.5
        span = rng.integers(nx // 4, nx // 2)
        x0 = rng.integers(border_width, nx - border_width - span)
        y0 = rng.integers(border_width + thick, ny - border_width - thick)
        if horizontal:
            img[y0 : y0 + thick, x0 : x0 + span] = 1.0
        else:
            img[x0 : x0 + span, y0 : y0 + thick] = 1.0
    return img


def synthetic_waypoints(
    img: np.ndarray,
    num_nodes: int,
    points_per_path: int = 8,
    seed: int = 0,
):
    """Random loop waypoints through free space, one array per node.

    Waypoints are in the reference's normalized [-1, 1]-ish convention
    (TrajectoryLidarDataset multiplies by nx/2, ny/2 —
    floorplans/lidar/lidar.py:311-318), closed (last == first) so cubic
    interpolation produces a loop. Each waypoint is rejection-sampled from
    free space.
    """
    rng = np.random.default_rng(seed)
    ny, nx = img.shape
    xs = nx * np.linspace(-0.5, 0.5, num=nx)
    ys = ny * np.linspace(-0.5, 0.5, num=ny)

    def is_free(xn, yn):
        # xn, yn are normalized in [-1, 1]; map to pixel indices
        ix = int(np.clip((xn * 0.5 + 0.5) * (nx - 1), 0, nx - 1))
        iy = int(np.clip((yn * 0.5 + 0.5) * (ny - 1), 0, ny - 1))
        return img[iy, ix] < 0.5

    del xs, ys

    def path_is_free(wp):
        """Whole interpolated trajectory must stay in free space (the
        spline can cut through walls between free waypoints)."""
        from .lidar import interpolate_waypoints

        traj = interpolate_waypoints(wp[:, 0], wp[:, 1], 8)
        return all(is_free(x, y) for x, y in traj)

    paths = []
    for _ in range(num_nodes):
        wp = None
        for _attempt in range(500):
            cx, cy = rng.uniform(-0.45, 0.45, size=2)
            r0 = rng.uniform(0.08, 0.25)
            angles = np.linspace(
                0, 2 * np.pi, points_per_path, endpoint=False
            )
            radii = r0 * rng.uniform(0.8, 1.05, size=points_per_path)
            px = np.clip(cx + radii * np.cos(angles), -0.8, 0.8)
            py = np.clip(cy + radii * np.sin(angles), -0.8, 0.8)
            cand = np.stack(
                [np.append(px, px[0]), np.append(py, py[0])], axis=1
            )
            if path_is_free(cand):
                wp = cand
                break
        if wp is None:
            raise RuntimeError(
                "could not place a wall-free trajectory; reduce "
                "num_walls or wall density"
            )
        paths.append(wp)
    return paths
