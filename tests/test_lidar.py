import numpy as np
import pytest
import torch

from nn_distributed_training_amd.data.floorplan import (
    synthetic_floorplan,
    synthetic_waypoints,
)
from nn_distributed_training_amd.data.lidar import (
    Lidar2D,
    OnlineTrajectoryLidarDataset,
    RandomPoseLidarDataset,
    TrajectoryLidarDataset,
    interpolate_waypoints,
)


@pytest.fixture(scope="module")
def lidar():
    img = synthetic_floorplan(nx=128, ny=128, num_walls=5, border_width=10,
                              seed=0)
    return Lidar2D(img, num_beams=8, beam_length=0.3, beam_samps=10,
                   samp_distribution_factor=1.0, collision_samps=30,
                   fine_samps=3)


def test_floorplan_structure():
    img = synthetic_floorplan(nx=64, ny=64, border_width=8, seed=1)
    assert img.shape == (64, 64)
    assert img[:8].min() == 1.0  # border walls
    assert img.min() == 0.0  # free space exists


def test_scan_shape_and_range(lidar):
    np.random.seed(0)
    scan = lidar.scan(np.zeros((1, 2)))
    # fixed samples per beam (beams that hit walls are refine-resampled)
    assert scan.shape == (8 * 10, 3)
    assert np.isfinite(scan).all()
    # first sample of each beam is at the scan origin with ~zero density
    starts = scan[::10]
    assert np.allclose(starts[:, :2], 0.0, atol=1e-9)
    assert (starts[:, 2] < 0.5).all()


def test_scan_from_wall_raises(lidar):
    # a point deep inside the border wall has density ~1
    with pytest.raises(NameError):
        lidar.scan(np.array([[lidar.xs[2], 0.0]]))


def test_scan_hits_walls(lidar):
    """Beams toward the border must terminate near a wall: last sample
    of some beam should have high density."""
    scan = lidar.scan(np.zeros((1, 2)))
    ends = scan[9::10]  # last sample per beam
    assert (ends[:, 2] >= 0.5).any()


def test_random_pose_dataset(lidar):
    np.random.seed(1)
    ds = RandomPoseLidarDataset(lidar, num_scans=5)
    assert len(ds) == 5 * 8 * 10
    x, y = ds[0]
    assert x.shape == (2,) and y.ndim == 0
    assert set(torch.unique(ds.scans[:, 2]).tolist()) <= {0.0, 1.0}


def test_trajectory_dataset(lidar):
    wps = synthetic_waypoints(lidar.img, 1, seed=3)[0]
    ds = TrajectoryLidarDataset(lidar, wps, spline_res=2)
    assert len(ds) > 0
    assert ds.scan_locs.shape[1] == 2


def test_online_dataset_window_advances(lidar):
    wps = synthetic_waypoints(lidar.img, 1, seed=4)[0]
    ds = OnlineTrajectoryLidarDataset(
        lidar, wps, spline_res=2, num_scans_in_window=3
    )
    pos0 = ds.curr_pos.copy()
    window = len(ds.curr_index_pool())
    assert window == 3 * 8 * 10
    # drain the window; the robot must advance
    for _ in range(window + 1):
        ds[0]
    assert not np.allclose(ds.curr_pos, pos0)


def test_interpolate_waypoints():
    x = np.array([0.0, 1.0, 2.0, 3.0])
    y = np.array([0.0, 1.0, 0.0, -1.0])
    out = interpolate_waypoints(x, y, spline_res=5)
    assert out.shape == (15, 2)
    # passes through the first waypoint
    assert np.allclose(out[0], [0.0, 0.0])


def test_online_window_sampler_fast_path_and_wrap():
    """_OnlineWindowSampler v2: synchronized equal windows take the
    zero-copy fast path (returns the pool tensor itself with stride =
    cap); window wraps advance the dataset and keep every returned
    index inside the node's current/previous window bounds."""
    import numpy as np
    import torch

    from nn_distributed_training_amd.data.floorplan import (
        synthetic_floorplan,
        synthetic_waypoints,
    )
    from nn_distributed_training_amd.data.lidar import (
        Lidar2D,
        OnlineTrajectoryLidarDataset,
    )
    from nn_distributed_training_amd.ops.stacked import (
        _OnlineWindowSampler,
    )

    np.random.seed(0)
    img = synthetic_floorplan(nx=96, ny=96, num_walls=3,
                              border_width=10, seed=0)
    lidar = Lidar2D(img, 6, 0.25, 8, 1.0, 20, 3)
    wps = synthetic_waypoints(img, 2, seed=0)
    dss = [
        OnlineTrajectoryLidarDataset(lidar, wp, 10, 4) for wp in wps
    ]
    win = dss[0].window_bounds[1] - dss[0].window_bounds[0]
    assert win == dss[1].window_bounds[1] - dss[1].window_bounds[0]
    B = max(1, win // 3)
    epochs = []
    s = _OnlineWindowSampler(dss, B, torch.device("cpu"), 0,
                             epochs.append)

    # fast path: the pool tensor itself comes back, no copy
    idx, stride, off = s.next_ref()
    assert idx is s.pools and stride == s.cap and off == 0
    idx2, stride2, off2 = s.next_ref()
    assert idx2 is s.pools and off2 == B

    # run through several windows; every index must stay in-bounds
    for _ in range(12):
        idx, stride, off = s.next_ref()
        for li in range(2):
            row = idx[li, off : off + B] if idx.shape[1] >= off + B \
                else idx[li]
            assert row.numel() == B
            assert int(row.min()) >= 0
            assert int(row.max()) < len(dss[li].tds)

    # determinism: same construction gives the same stream
    np.random.seed(0)
    dss2 = [
        OnlineTrajectoryLidarDataset(lidar, wp, 10, 4) for wp in wps
    ]
    s2 = _OnlineWindowSampler(dss2, B, torch.device("cpu"), 0,
                              lambda li: None)
