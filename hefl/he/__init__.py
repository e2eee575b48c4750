from .ckks import CKKSContext, Ciphertext, CtxtTensor, KeyPair
from .pyfhel_compat import Pyfhel, PyCtxt

__all__ = ["CKKSContext", "Ciphertext", "CtxtTensor", "KeyPair", "Pyfhel", "PyCtxt"]
