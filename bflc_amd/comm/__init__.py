from bflc_amd.comm.transport import Transport

__all__ = ["Transport"]
