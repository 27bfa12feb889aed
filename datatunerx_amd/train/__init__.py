from .args import DataArguments, FinetuningArguments, ModelArguments
from .trainer import SFTTrainer, TrainerConfig
