"""Loaders package (parity: reference ``veles/loader/``)."""

from veles.loader.base import (  # noqa: F401
    Loader, ILoader, TEST, VALID, TRAIN, CLASS_NAME)
from veles.loader.fullbatch import (  # noqa: F401
    FullBatchLoader, IFullBatchLoader, FullBatchLoaderMSE)
