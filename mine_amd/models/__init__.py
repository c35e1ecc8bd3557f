from mine_amd.models.resnet import ResNetEncoder  # noqa: F401
from mine_amd.models.decoder import MPIDecoder  # noqa: F401
from mine_amd.models.vdr import VDRPredictor  # noqa: F401
