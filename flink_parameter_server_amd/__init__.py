"""MI355X-native asynchronous parameter-server training framework.

A from-scratch re-design of ``FlinkML/flink-parameter-server`` for AMD
Instinct MI355X: the reference's Flink iterative-stream loop between
workers and parameter shards becomes round-based RCCL send/recv over xGMI
(one process per GPU, each hosting one worker and one server shard whose
partition lives dense in HBM3E); the server-side sparse applies and the
worker-side model steps are hand-written HIP/CDNA4 kernels.

See SURVEY.md for the reference's structure and the design mapping.
"""
__version__ = "0.2.0"

from .api import (ParameterServer, ParameterServerClient, ServerLogic,
                  SimpleServerLogic, WorkerLogic, add_pull_limiter)
from .config import PSConfig
from .table import ShardedTable, TableServerLogic
from .parallel.runtime import PSEngine
from .parallel.transport import (LocalTransport, LoopbackTransport,
                                 TorchDistTransport, Transport)
from .parallel.partitioner import (ModPartitioner, RangePartitioner,
                                   make_partitioner)
from .transform import (transform, transform_distributed,
                        transform_local, transform_simple,
                        transform_with_model_load)
from .messaging import (CountFlush, FlushAlways, FlushStrategy, Pull,
                        PullAnswer, Push, PSToWorker, SerializingTransport,
                        TimeFlush, WorkerToPS)
from .algorithms import (ps_offline_mf, ps_online_mf, run_count_min,
                         train_lr, train_sgns, transform_binary,
                         transform_multiclass)
from .models.sketch import (AMSSketchWorkerLogic, BloomFilterWorkerLogic,
                            CountMinWorkerLogic)
from .utils.checkpoint import (PeriodicCheckpointer, latest_checkpoint,
                               load_shard, save_shard)
