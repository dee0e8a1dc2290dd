"""dpark_amd — MI355X-native shuffle/sort/group-by engine behind the
douban/dpark RDD API surface.

A brand-new build (NOT a port): the API layer mirrors dpark's
DparkContext / RDD surface for the hot path (reduceByKey, groupByKey,
combineByKey, sortByKey, join, cogroup — SURVEY.md §2 rows 1-3) so it is
a drop-in for that path; the shuffle engine underneath
(dpark/shuffle.py's LocalFileShuffle + Merger + ShuffleFetcher [U]) is
replaced by hand-written HIP/gfx950 kernels behind a ctypes C-ABI
(include/dprk.h, dpark_amd/csrc/) with the exchange as one RCCL
all-to-allv.  Generic Python-object RDDs run on the CPU local scheduler
(that IS dpark's behavior for them); int64/float64 pair RDDs route to the
GPU engine, which FAILS LOUDLY if its extension is missing on a GPU box
(no silent CPU fallback on the engine path).
"""

from .context import DparkContext
from .rdd import RDD

__version__ = "0.1.0"
__all__ = ["DparkContext", "RDD"]
