from bflc_amd.utils.metrics import JsonlLogger

__all__ = ["JsonlLogger"]
