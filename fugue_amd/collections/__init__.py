from fugue_amd.collections.partition import (
    BagPartitionCursor,
    DatasetPartitionCursor,
    PartitionCursor,
    PartitionSpec,
    parse_presort_exp,
)
from fugue_amd.collections.sql import StructuredRawSQL, TempTableName
from fugue_amd.collections.yielded import PhysicalYielded, Yielded
