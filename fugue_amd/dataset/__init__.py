from fugue_amd.dataset.dataset import Dataset, DatasetDisplay, get_dataset_display
from fugue_amd.dataset.api import (
    as_fugue_dataset,
    count,
    is_bounded,
    is_empty,
    is_local,
    show,
)
from fugue_amd.dataset.dataset import AnyDataset  # noqa: E402
