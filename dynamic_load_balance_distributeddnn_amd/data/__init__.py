from .corpus import Corpus, Dictionary, corpus_available  # noqa: F401
from .partition import (GlobalBatchStream, batchify, bptt_batch,  # noqa: F401
                        partition_cv, partition_lm)
from .synthetic import DATASET_SPECS, make_cv_dataset, make_lm_tokens  # noqa: F401
from .real import load_cv_dataset, load_lm_tokens  # noqa: F401
