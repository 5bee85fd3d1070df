from .partitioner import ModPartitioner, Partitioner, RangePartitioner, bucket_by_shard, make_partitioner
from .runtime import Metrics, PSEngine
from .transport import LocalTransport, LoopbackTransport, TorchDistTransport, Transport
