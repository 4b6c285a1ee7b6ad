from .distributed import (setup_distributed, teardown_distributed,  # noqa: F401
                          auto_backend, bind_gpu, DistContext)
from .launcher import run_workers, aggregate_worker_csvs  # noqa: F401
