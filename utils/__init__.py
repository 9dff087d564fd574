"""Compatibility façade: the reference exposes its API as the ``utils``
package (reference utils/__init__.py:1-5).  Code written against the
reference keeps working against this framework unchanged:

    from utils import same_seeds, EarlyStopping, Lookahead, Toy_Net, \
        iterate_loader, train
"""
from ddp_tricks_amd import (  # noqa: F401
    same_seeds, EarlyStopping, Lookahead, Toy_Net, iterate_loader, train,
)
