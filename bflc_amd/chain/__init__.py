from bflc_amd.chain import records
from bflc_amd.chain.client import CONTRACT_ADDRESS, BcosClient
from bflc_amd.chain.local_chain import LocalChain

__all__ = ["records", "BcosClient", "LocalChain", "CONTRACT_ADDRESS"]
