from .weights import flat_params, load_flat_params
from .aggregate import plaintext_fedavg
from .client import LocalClient

__all__ = ["flat_params", "load_flat_params", "plaintext_fedavg", "LocalClient"]
