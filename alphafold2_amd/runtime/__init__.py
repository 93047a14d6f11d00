from .graphed import GraphedTrainStep
from .checkpointing import save_checkpoint, load_checkpoint
