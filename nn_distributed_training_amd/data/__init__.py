from .mnist import load_mnist, split_train_set  # noqa: F401
from .lidar import (  # noqa: F401
    Lidar2D,
    OnlineTrajectoryLidarDataset,
    RandomPoseLidarDataset,
    TrajectoryLidarDataset,
    interpolate_waypoints,
)
from .floorplan import synthetic_floorplan, synthetic_waypoints  # noqa: F401
