from .mf import MFBatch, OnlineMFWorkerLogic, epoch_iterator
from .pa import PAWorkerLogic, SparseBatch
from .lr import LRWorkerLogic
from .sgns import SGNSBatch, SGNSWorkerLogic
