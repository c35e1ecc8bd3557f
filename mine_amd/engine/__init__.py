from mine_amd.engine.task import SynthesisTask  # noqa: F401
from mine_amd.engine.checkpoint import save_checkpoint, restore_model  # noqa: F401
