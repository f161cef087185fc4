from relora_amd.data.dataloader import (  # noqa: F401
    PreprocessedIterableDataset,
    SkipBatchSampler,
    SkipDataLoader,
    SyntheticDataset,
    tokenize_and_chunk,
)
