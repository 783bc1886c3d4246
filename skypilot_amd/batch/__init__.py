from skypilot_amd.batch.core import Dataset, run_batch

__all__ = ["Dataset", "run_batch"]
