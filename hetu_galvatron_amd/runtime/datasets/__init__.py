from .indexed import IndexedDataset, IndexedDatasetBuilder  # noqa: F401
from .gpt_dataset import (  # noqa: F401
    BlendedDataset, GPTDataset, build_pretraining_dataset)
