from .indexed import (IndexedDataset, IndexedDatasetBuilder,  # noqa: F401
                      MegatronIndexedDataset, MegatronIndexedDatasetBuilder,
                      load_indexed_dataset)
from .gpt_dataset import (  # noqa: F401
    BlendedDataset, GPTDataset, build_pretraining_dataset)
