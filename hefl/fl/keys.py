"""Key-file workflow — API parity with the reference's key management
(FLPyfhelin.py:330-364 gen_pk/get_pk/get_sk/gen_rekey, notebook cell 1).

The reference pickles {HE object, context bytes, pk bytes} into
publickey.pickle and additionally the secret key into privatekey.pickle
(notebook cell 1). Same files, same separation: the aggregator loads only
publickey.pickle; the decrypting party loads privatekey.pickle. The
reference's gen_rekey is dead code with a NameError (:363); ours works.
"""
from __future__ import annotations

import os
import pickle
from typing import Optional

from ..he.pyfhel_compat import Pyfhel

PUBLIC_KEY_FILE = "publickey.pickle"
PRIVATE_KEY_FILE = "privatekey.pickle"


def gen_pk(s: int = 128, m: int = 2048, directory: str = ".",
           scale_bits: int = 40, q_bits=(60, 40),
           seed: Optional[int] = None) -> Pyfhel:
    """Generate context + keypair; export public material to
    publickey.pickle and the secret key to privatekey.pickle
    (reference gen_pk FLPyfhelin.py:330-344 + notebook cell 1)."""
    HE = Pyfhel()
    HE.contextGen(p=65537, m=m, sec=s, scale_bits=scale_bits, q_bits=q_bits,
                  seed=seed)
    HE.keyGen()
    os.makedirs(directory, exist_ok=True)
    with open(os.path.join(directory, PUBLIC_KEY_FILE), "wb") as f:
        pickle.dump({"HE": HE, "con": HE.to_bytes_context(),
                     "pk": HE.to_bytes_publicKey()}, f,
                    protocol=pickle.HIGHEST_PROTOCOL)
    with open(os.path.join(directory, PRIVATE_KEY_FILE), "wb") as f:
        pickle.dump({"con": HE.to_bytes_context(),
                     "pk": HE.to_bytes_publicKey(),
                     "sk": HE.to_bytes_secretKey()}, f,
                    protocol=pickle.HIGHEST_PROTOCOL)
    return HE


def get_pk(directory: str = ".") -> Pyfhel:
    """Restore context + public key only (reference get_pk
    FLPyfhelin.py:346-355) — what the aggregation server holds (:370)."""
    with open(os.path.join(directory, PUBLIC_KEY_FILE), "rb") as f:
        d = pickle.load(f)
    HE = Pyfhel()
    HE.from_bytes_context(d["con"])
    HE.from_bytes_publicKey(d["pk"])
    return HE


def get_sk(directory: str = ".") -> Pyfhel:
    """Restore context + pk + SECRET key (reference get_sk
    FLPyfhelin.py:251-261) — the decrypting party only."""
    with open(os.path.join(directory, PRIVATE_KEY_FILE), "rb") as f:
        d = pickle.load(f)
    HE = Pyfhel()
    HE.from_bytes_context(d["con"])
    HE.from_bytes_publicKey(d["pk"])
    HE.from_bytes_secretKey(d["sk"])
    return HE


def gen_rekey(HE: Pyfhel, bitCount: int = 1, size: int = 5) -> Pyfhel:
    """Relinearization keygen (reference gen_rekey FLPyfhelin.py:357-364 —
    dead code raising NameError there; functional here)."""
    HE.relinKeyGen(bitCount=bitCount, size=size)
    return HE
