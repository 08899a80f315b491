from .data import Partition, DataPartitioner, SyntheticMNIST, partition_dataset  # noqa: F401
from .ddp import average_gradients, DistributedDataParallel  # noqa: F401
