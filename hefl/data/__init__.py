from .synthetic import SyntheticMedicalImages, make_client_loader
from .shard import shard_indices, prep_df

__all__ = ["SyntheticMedicalImages", "make_client_loader", "shard_indices", "prep_df"]
