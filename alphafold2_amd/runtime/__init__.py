from .graphed import GraphedTrainStep
from .checkpointing import (save_checkpoint, load_checkpoint,
                            convert_trunk_state_dict)
from .profiling import StepTimer, kernel_stats_summary, profile_trace
from .tuning import enable_tuned_gemm
