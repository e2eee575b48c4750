from .aggregate import plaintext_fedavg
from .client import LocalClient
from .round import FLRunner
from .secure import SecureAggregator
from .sequential import SequentialFL, train_server
from .weights import flat_params, load_flat_params

__all__ = ["flat_params", "load_flat_params", "plaintext_fedavg",
           "LocalClient", "FLRunner", "SecureAggregator", "SequentialFL",
           "train_server"]
