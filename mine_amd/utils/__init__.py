from mine_amd.utils.misc import (  # noqa: F401
    AverageMeter,
    disparity_normalization_vis,
    linspace_batch,
    setup_logger,
)
from mine_amd.utils.geometry import inverse_3x3, inverse_rigid_4x4, inverse_4x4  # noqa: F401
from mine_amd.utils.embedder import PositionalEncoder, get_embedder  # noqa: F401
