from .abstractbasedataset import AbstractBaseDataset, dataset_name_to_id
from .pickledataset import SimplePickleDataset, SimplePickleWriter
from .abstractrawdataset import AbstractRawDataset
from .rawloaders import LSMSDataset, XYZDataset, CFGDataset
from .graphstore import ShardedDistDataset, GraphStoreWriter, GraphStoreDataset, DistDataset
from .serializeddataset import SerializedDataset, SerializedWriter
from .adios_compat import AdiosWriter, AdiosDataset
