"""2-D lidar simulator + occupancy datasets for the density workloads.

Behavioral parity with the reference's ``floorplans/lidar/lidar.py``:
a spline-interpolated density image is ray-marched with coarse collision
sampling and fine refinement, producing (x, y, density) training tuples;
datasets provide random-pose scans, full-trajectory scans, and an *online*
sliding-window variant whose ``curr_pos`` drives the dynamic communication
graph (problems/dist_online_dense_problem.py).

Unlike the reference (per-beam Python loops, lidar.py:84-135), the scan is
vectorized over beams. The source image can be a PNG path or any ndarray
(see data/floorplan.py for procedural floorplans).
"""

from __future__ import annotations

import random

import numpy as np
import scipy.interpolate as interp
import torch


def _load_img(img, border_width: int) -> np.ndarray:
    if isinstance(img, (str,)):
        from PIL import Image

        img = np.asarray(Image.open(img)).astype(float) / 255.0
    else:
        img = np.asarray(img, dtype=float)
    if border_width != 0:
        img = img.copy()
        img[:, :border_width] = 1.0
        img[:border_width, :] = 1.0
        # `-1` end quirk kept for parity: the reference leaves the very
        # last row/column unbordered (floorplans/lidar/lidar.py:39-42)
        img[:, -border_width:-1] = 1.0
        img[-border_width:-1, :] = 1.0
    return img


class Lidar2D:
    """Queryable 2-D lidar over a density image.

    ``scan(pos)`` returns an array [num_beams * beam_samps, 3] of
    (x, y, density) samples: beams that hit a wall are fine-refined and
    re-sampled toward the collision point with density-weighted spacing
    (``samp_distribution_factor``); unobstructed beams sample uniformly.
    """

    def __init__(
        self,
        img,
        num_beams: int,
        beam_length: float,
        beam_samps: int,
        samp_distribution_factor: float,
        collision_samps: int,
        fine_samps: int,
        border_width: int = 0,
    ):
        self.img = _load_img(img, border_width)
        self.beam_stop_thresh = 0.5
        self.num_beams = num_beams
        self.beam_samps = beam_samps
        self.collision_samps = collision_samps
        self.fine_samps = fine_samps
        self.samp_df = samp_distribution_factor

        self.nx = self.img.shape[1]
        self.ny = self.img.shape[0]
        self.beam_len = beam_length * max(self.nx, self.ny)
        self.xs = self.nx * np.linspace(-0.5, 0.5, num=self.nx)
        self.ys = self.ny * np.linspace(-0.5, 0.5, num=self.ny)
        self.density = interp.RectBivariateSpline(
            self.xs, self.ys, self.img.T
        )

    # ------------------------------------------------------------------
    def scan(self, pos: np.ndarray) -> np.ndarray:
        pos = np.asarray(pos, dtype=float).reshape(1, 2)
        if self.density.ev(pos[0, 0], pos[0, 1]) >= self.beam_stop_thresh:
            raise NameError("Cannot lidar scan from point with high density.")

        B, S = self.num_beams, self.collision_samps
        angs = np.linspace(-np.pi, np.pi, num=B, endpoint=False)
        beam_vec = self.beam_len * np.stack(
            [np.cos(angs), np.sin(angs)], axis=1
        )  # [B, 2]

        # coarse march all beams at once
        t = np.linspace(0.0, 1.0, num=S).reshape(1, S, 1)
        coarse = pos.reshape(1, 1, 2) + t * beam_vec.reshape(B, 1, 2)
        vals = self.density.ev(
            coarse[..., 0].ravel(), coarse[..., 1].ravel()
        ).reshape(B, S)
        hits = vals >= self.beam_stop_thresh
        hit_idx = np.argmax(hits, axis=1)  # 0 when no hit (parity w/ ref)

        # endpoint per beam: free beams end at full length; hit beams end at
        # the fine-refined collision point
        endpoints = pos + beam_vec  # [B, 2]
        hit_beams = np.nonzero(hit_idx > 0)[0]
        if hit_beams.size:
            F = self.fine_samps
            last_empty = coarse[hit_beams, hit_idx[hit_beams] - 1]  # [H,2]
            first_hit = coarse[hit_beams, hit_idx[hit_beams]]
            tf = np.linspace(0.0, 1.0, num=F).reshape(1, F, 1)
            fine = last_empty[:, None, :] + tf * (
                first_hit - last_empty
            )[:, None, :]
            fvals = self.density.ev(
                fine[..., 0].ravel(), fine[..., 1].ravel()
            ).reshape(-1, F)
            fidx = np.argmax(fvals >= self.beam_stop_thresh, axis=1)
            endpoints[hit_beams] = fine[np.arange(len(hit_beams)), fidx]

        # sample points along each beam: uniform spacing for free beams,
        # density-weighted (t^samp_df) toward the wall for hit beams
        M = self.beam_samps
        tu = np.linspace(0.0, 1.0, M)
        tw = np.power(tu, self.samp_df)
        t_all = np.where((hit_idx > 0)[:, None], tw[None, :], tu[None, :])
        pnts = pos.reshape(1, 1, 2) + t_all[..., None] * (
            endpoints - pos
        ).reshape(B, 1, 2)
        svals = self.density.ev(
            pnts[..., 0].ravel(), pnts[..., 1].ravel()
        ).reshape(B, M, 1)
        return np.concatenate([pnts, svals], axis=2).reshape(B * M, 3)


class ClippedLidar2D:
    """Variable-length scans: beams stop at the first collision sample.

    Parity with reference floorplans/lidar/lidar.py:139-237.
    """

    def __init__(self, img, num_beams, beam_length, beam_samps,
                 border_width=0):
        self.img = _load_img(img, border_width)
        self.beam_stop_thresh = 0.5
        self.num_beams = num_beams
        self.beam_samps = beam_samps
        self.nx = self.img.shape[1]
        self.ny = self.img.shape[0]
        self.beam_len = beam_length * max(self.nx, self.ny)
        self.xs = self.nx * np.linspace(-0.5, 0.5, num=self.nx)
        self.ys = self.ny * np.linspace(-0.5, 0.5, num=self.ny)
        self.density = interp.RectBivariateSpline(self.xs, self.ys, self.img.T)

    def scan(self, pos):
        pos = np.asarray(pos, dtype=float).reshape(1, 2)
        if self.density.ev(pos[0, 0], pos[0, 1]) >= self.beam_stop_thresh:
            raise NameError("Cannot lidar scan from point with high density.")
        B, S = self.num_beams, self.beam_samps
        angs = np.linspace(-np.pi, np.pi, num=B, endpoint=False)
        beam_vec = self.beam_len * np.stack(
            [np.cos(angs), np.sin(angs)], axis=1
        )
        t = np.linspace(0.0, 1.0, num=S).reshape(1, S, 1)
        pnts = pos.reshape(1, 1, 2) + t * beam_vec.reshape(B, 1, 2)
        vals = self.density.ev(
            pnts[..., 0].ravel(), pnts[..., 1].ravel()
        ).reshape(B, S)
        out = []
        for b in range(B):
            hit = np.argmax(vals[b] >= self.beam_stop_thresh)
            end = S if hit == 0 else hit + 1
            out.append(
                np.concatenate(
                    [pnts[b, :end], vals[b, :end, None]], axis=1
                )
            )
        return np.vstack(out)


# ----------------------------------------------------------------------
class RandomPoseLidarDataset(torch.utils.data.Dataset):
    """Scans from poses rejection-sampled in free space (validation data).

    Parity with reference floorplans/lidar/lidar.py:240-287.
    """

    def __init__(self, lidar: Lidar2D, num_scans: int, round_density=True):
        super().__init__()
        self.lidar = lidar
        locs, c = [], 0
        while c < num_scans:
            xsamp = np.random.choice(lidar.xs, num_scans)
            ysamp = np.random.choice(lidar.ys, num_scans)
            mask = lidar.density.ev(xsamp, ysamp) < 0.5
            c += int(mask.sum())
            locs.append(np.stack([xsamp[mask], ysamp[mask]], axis=1))
        self.scan_locs = np.vstack(locs)[:num_scans]
        scans = np.vstack(
            [lidar.scan(self.scan_locs[k]) for k in range(num_scans)]
        )
        self.scans = torch.from_numpy(scans).to(torch.get_default_dtype())
        if round_density:
            self.scans[:, 2] = torch.round(self.scans[:, 2])
        self.tds = torch.utils.data.TensorDataset(
            self.scans[:, :2], self.scans[:, 2]
        )

    def __getitem__(self, idx):
        return self.tds[idx]

    def __len__(self):
        return len(self.tds)


class TrajectoryLidarDataset(torch.utils.data.Dataset):
    """All scans along a spline-interpolated waypoint trajectory.

    Parity with reference floorplans/lidar/lidar.py:290-333.
    """

    def __init__(self, lidar, waypoints, spline_res, round_density=True):
        super().__init__()
        self.lidar = lidar
        traj = interpolate_waypoints(
            waypoints[:, 0], waypoints[:, 1], spline_res
        )
        conv = np.array([lidar.nx * 0.5, lidar.ny * 0.5]).reshape(1, 2)
        self.scan_locs = traj * conv
        scans = np.vstack(
            [lidar.scan(self.scan_locs[k]) for k in range(len(traj))]
        )
        self.scans = torch.from_numpy(scans).to(torch.get_default_dtype())
        if round_density:
            self.scans[:, 2] = torch.round(self.scans[:, 2])
        self.tds = torch.utils.data.TensorDataset(
            self.scans[:, :2], self.scans[:, 2]
        )

    def __getitem__(self, idx):
        return self.tds[idx]

    def __len__(self):
        return len(self.tds)


class OnlineTrajectoryLidarDataset(torch.utils.data.Dataset):
    """Sliding-window streaming dataset over a robot trajectory.

    The robot advances ``num_scans_in_window`` scans each time the window
    is exhausted; ``curr_pos`` exposes the robot's current trajectory
    position (consumed by the dynamic-graph problem every round,
    problems/dist_online_dense_problem.py:141-155 in the reference).
    Windows wrap around at the end of the trajectory.
    """

    def __init__(
        self, lidar, waypoints, spline_res, num_scans_in_window,
        round_density=True,
    ):
        super().__init__()
        self.lidar = lidar
        traj = interpolate_waypoints(
            waypoints[:, 0], waypoints[:, 1], spline_res
        )
        self.num_scans = traj.shape[0]
        conv = np.array([lidar.nx * 0.5, lidar.ny * 0.5]).reshape(1, 2)
        self.scan_locs = traj * conv
        scans = np.vstack(
            [lidar.scan(self.scan_locs[k]) for k in range(self.num_scans)]
        )
        self.scans = torch.from_numpy(scans).to(torch.get_default_dtype())
        if round_density:
            self.scans[:, 2] = torch.round(self.scans[:, 2])
        self.tds = torch.utils.data.TensorDataset(
            self.scans[:, :2], self.scans[:, 2]
        )

        self.num_scans_in_window = num_scans_in_window
        self.scan_size = lidar.num_beams * lidar.beam_samps
        self.curr_scan_idx = 0
        self.curr_pos = self.scan_locs[0]
        self.curr_idx_list: list = []
        # dataset-local RNG: window shuffles must not depend on global
        # random-module state (which diverges with rank packing — the
        # same node must draw the same stream on any rank)
        self._shuffle_rng = random.Random(0xD5)
        self._advance_window()

    def _advance_window(self):
        """Move the window forward one step; index pool refills lazily
        (the stacked engine samples the window's [lb, ub) bounds with a
        device-side randperm instead of a host list shuffle)."""
        w = self.num_scans_in_window
        start = self.curr_scan_idx
        end = min(start + w, self.num_scans)
        self.curr_scan_idx = end % self.num_scans if end >= self.num_scans \
            else end
        self.window_bounds = (self.scan_size * start, self.scan_size * end)
        self.curr_pos = self.scan_locs[min(end, self.num_scans - 1)]
        self.curr_idx_list = None  # lazy; see curr_index_pool()

    def curr_index_pool(self):
        if self.curr_idx_list is None:
            lb, ub = self.window_bounds
            self.curr_idx_list = list(range(lb, ub))
            self._shuffle_rng.shuffle(self.curr_idx_list)
        return self.curr_idx_list

    # the reference pops shuffled indices until the window empties, then
    # slides the window (lidar.py:383-424); same contract here
    def __getitem__(self, index):
        if not self.curr_index_pool():
            self._advance_window()
        return self.tds[self.curr_index_pool().pop()]

    def __len__(self):
        return len(self.tds)


def interpolate_waypoints(x, y, spline_res):
    """Cubic spline through waypoints; parity with lidar.py:427-435."""
    i = np.arange(len(x))
    interp_i = np.linspace(0, i.max(), spline_res * i.max())
    xi = interp.interp1d(i, x, kind="cubic")(interp_i)
    yi = interp.interp1d(i, y, kind="cubic")(interp_i)
    return np.stack([xi, yi], axis=1)
