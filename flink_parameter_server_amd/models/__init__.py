from .mf import (MFBatch, OnlineMFWorkerLogic, Rating,
                 SGDUpdater, epoch_iterator)
from .pa import PAWorkerLogic, SparseBatch
from .lr import LRWorkerLogic
from .sgns import SGNSBatch, SGNSWorkerLogic
