"""asyncframework_amd — MI355X-native asynchronous optimization engine.

A from-scratch rebuild of the capabilities of ASYNCframework/ASYNCframework
(IPDPS 2020 "ASYNC": bounded-staleness asynchronous ASGD/ASAGA on Spark)
as a single-node multi-GPU parameter-server runtime:

* one process per GPU over torch.distributed (RCCL over xGMI),
* a driver-side mailbox (``ASYNCcontext``) + worker-state table, API-compatible
  with the reference (reference: core/src/main/scala/org/apache/spark/rdd/
  ASYNCcontext.scala:14-81),
* hand-written HIP/CDNA4 kernels for the gradient / SAGA-history / update hot
  path (see ``csrc/``),
* the five algorithm drivers (ASGD/ASAGA x async/sync + sync mini-batch SGD
  baseline) with the reference's 13-positional-arg CLI and stdout contract.
"""

__version__ = "0.1.0"

from .core.context import ASYNCcontext, RDDPartialRes, workerState  # noqa: F401
from .core.rdd import AsyncRDD, ASYNCbroadcast  # noqa: F401
