from .corpus import Corpus, Dictionary, corpus_available  # noqa: F401
from .partition import batchify, bptt_batch, partition_cv, partition_lm  # noqa: F401
from .synthetic import DATASET_SPECS, make_cv_dataset, make_lm_tokens  # noqa: F401
from .real import load_cv_dataset, load_lm_tokens  # noqa: F401
