from mine_amd.parallel.ddp import GradAllReduceEngine, init_distributed  # noqa: F401
