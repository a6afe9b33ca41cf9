from .dist_helpers import run_distributed, find_free_port  # noqa: F401
