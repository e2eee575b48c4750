from .files import FileImageDataset
from .pipeline import get_test_data, get_train_data, hflip_augment
from .shard import prep_df, shard_indices
from .synthetic import (ClientLoader, SyntheticMedicalImages, affine_sample,
                        make_client_loader)

__all__ = ["SyntheticMedicalImages", "ClientLoader", "make_client_loader",
           "FileImageDataset", "shard_indices", "prep_df", "get_train_data",
           "get_test_data", "hflip_augment", "affine_sample"]
