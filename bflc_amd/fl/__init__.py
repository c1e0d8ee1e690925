from bflc_amd.fl.engine import (FLEngine, RoundStats, StragglerError,
                                client_rank)

__all__ = ["FLEngine", "RoundStats", "StragglerError", "client_rank"]
