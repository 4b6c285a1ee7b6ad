from .seed import seed_everything, shared_subset_indices  # noqa: F401
from .ports import find_free_port  # noqa: F401
