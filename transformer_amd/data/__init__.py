from .tokenizer import SubwordTokenizer  # noqa: F401
from .dataset import (  # noqa: F401
    read_data,
    load_dataset,
    load_or_create_tokenizer,
    SyntheticSeq2SeqDataset,
    BatchedDataset,
)
