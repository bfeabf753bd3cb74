"""dampr_amd — an MI355X-native out-of-core dataflow engine with the
capabilities and Python API of Dampr (Spark/Scalding-style MapReduce DSL).

The DSL and plan layers mirror the reference API.
Execution is tiered:

* CPU tier: multi-process out-of-core engine (runner.py / executor.py) —
  works everywhere, used for arbitrary Python UDFs.
* GPU tier (dampr_amd.gpu): columnar record batches in HBM3E with
  hand-written gfx950 HIP kernels for the shuffle/sort/combine/join core and
  RCCL over xGMI for the multi-GPU exchange.
"""
import logging

from .dampr import Dampr, PMap, PReduce, PJoin, ARReduce, ValueEmitter
from .base import BlockMapper, BlockReducer, Map, StreamMapper, \
    StreamReducer, Reduce
from .dataset import Dataset, Chunker
from . import funcs, settings

__all__ = ["Dampr", "PMap", "PReduce", "PJoin", "ARReduce", "ValueEmitter", "funcs",
           "BlockMapper", "BlockReducer", "Dataset", "Chunker", "settings",
           "setup_logging"]

__version__ = "0.1.0"


def setup_logging(debug=False):
    """Convenience logging setup."""
    loglevel = logging.DEBUG if debug else logging.INFO
    logging.basicConfig(level=loglevel,
                        format="%(asctime)s %(levelname)s %(message)s")
