from bflc_amd.fl.engine import FLEngine, RoundStats, client_rank

__all__ = ["FLEngine", "RoundStats", "client_rank"]
