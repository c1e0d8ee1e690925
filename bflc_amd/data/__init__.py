from bflc_amd.data.occupancy import (DEFAULT_PATH as OCCUPANCY_PATH,
                                     load_occupancy, occupancy_available)
from bflc_amd.data.synthetic import (Shard, make_federated, make_images,
                                     make_tabular, partition_dirichlet,
                                     partition_iid)

__all__ = ["Shard", "make_federated", "make_tabular", "make_images",
           "partition_iid", "partition_dirichlet", "load_occupancy",
           "occupancy_available", "OCCUPANCY_PATH"]
