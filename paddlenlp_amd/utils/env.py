"""Filename constants and env helpers (reference: paddlenlp/utils/env.py:68-110)."""
import os

CONFIG_NAME = "config.json"
GENERATION_CONFIG_NAME = "generation_config.json"

# model weights (safetensors is the only native format of this framework)
SAFE_WEIGHTS_NAME = "model.safetensors"
SAFE_WEIGHTS_INDEX_NAME = "model.safetensors.index.json"

# unified checkpoint (reference: paddlenlp/utils/env.py:97-110)
SAFE_OPTIMIZER_NAME = "optimizer.safetensors"
SAFE_OPTIMIZER_INDEX_NAME = "optimizer.safetensors.index.json"
SAFE_MASTER_WEIGHTS_NAME = "master_weights.safetensors"
SAFE_MASTER_WEIGHTS_INDEX_NAME = "master_weights.safetensors.index.json"

TRAINER_STATE_NAME = "trainer_state.json"
TRAINING_ARGS_NAME = "training_args.bin"
OPTIMIZER_STATE_NAME = "optimizer.pt"
SCHEDULER_NAME = "scheduler.pt"
SCALER_NAME = "scaler.pt"

TOKENIZER_CONFIG_NAME = "tokenizer_config.json"

PREFIX_CHECKPOINT_DIR = "checkpoint"


def get_cache_home():
    return os.environ.get(
        "PDNLP_AMD_HOME", os.path.join(os.path.expanduser("~"), ".paddlenlp_amd")
    )
