from .types import (Dataset, Finetune, FinetuneExperiment, FinetuneJob,
                    Hyperparameter, LLM, LLMCheckpoint, Scoring)
