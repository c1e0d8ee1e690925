from bflc_amd.data.synthetic import (Shard, make_federated, make_images,
                                     make_tabular, partition_dirichlet,
                                     partition_iid)

__all__ = ["Shard", "make_federated", "make_tabular", "make_images",
           "partition_iid", "partition_dirichlet"]
