"""TrainingArguments: distributed config + process-group bootstrap.

Reference behavior: paddlenlp/trainer/training_args.py:71 (the ~120-field
dataclass) and :887-1405 (__post_init__ validating degrees, deriving
data_parallel_degree and initializing the hybrid topology in the order
["dp","pp","sharding","sep","mp"]).  Here the topology is built directly as
torch.distributed process groups (RCCL over xGMI on a GPU node, gloo on CPU).
"""
from __future__ import annotations

import json
import os
from dataclasses import dataclass, field
from typing import List, Optional

import torch

from ..parallel.topology import Topology, get_topology, init_parallel_env
from ..utils.log import logger
from .trainer_utils import ShardingOption


@dataclass
class TrainingArguments:
    output_dir: str = field(default="output")
    overwrite_output_dir: bool = False

    do_train: bool = False
    do_eval: bool = False
    do_predict: bool = False

    per_device_train_batch_size: int = 8
    per_device_eval_batch_size: int = 8
    gradient_accumulation_steps: int = 1

    learning_rate: float = 5e-5
    weight_decay: float = 0.0
    adam_beta1: float = 0.9
    adam_beta2: float = 0.999
    adam_epsilon: float = 1e-8
    max_grad_norm: float = 1.0

    num_train_epochs: float = 1.0
    max_steps: int = -1
    lr_scheduler_type: str = "linear"
    warmup_ratio: float = 0.0
    warmup_steps: int = 0
    min_lr_ratio: float = 0.0

    logging_steps: int = 10
    logging_dir: Optional[str] = None
    save_strategy: str = "steps"
    logging_strategy: str = "steps"
    save_steps: int = 500
    save_total_limit: Optional[int] = None
    evaluation_strategy: str = "no"
    eval_steps: Optional[int] = None

    seed: int = 42

    # precision (bf16 O2 is the MI355X-native default path)
    bf16: bool = False
    fp16: bool = False
    amp_master_grad: bool = True

    # parallelism degrees (reference: training_args.py:554-875)
    tensor_parallel_degree: int = 1
    pipeline_parallel_degree: int = 1
    sharding_parallel_degree: int = -1
    sep_parallel_degree: int = 1
    context_parallel_degree: int = 1
    # zigzag load-balanced CP shards (rank r owns chunks r and 2w-1-r)
    context_parallel_balanced: bool = False
    sharding: str = ""  # "stage1" | "stage2" | "stage3" | "" (space-separated options)
    sharding_comm_buffer_size_MB: int = 256
    # overlap the sharding-group gradient reduction with backward compute
    sharding_overlap_comm: bool = True
    # interleaved virtual pipeline stages per rank (VPP; 1 = plain 1F1B)
    virtual_pp_degree: int = 1
    # [[start, end], ...] 1-based global-step intervals whose data is skipped
    # (corrupted-range replay jump, reference trainer.py should_skip_data)
    skip_data_intervals: Optional[List[List[int]]] = None
    skip_memory_metrics: bool = True
    use_expert_parallel: bool = False
    expert_parallel_degree: int = 1

    tensor_parallel_output: bool = True
    sequence_parallel: bool = False

    # recompute
    recompute: bool = False
    recompute_granularity: str = "full"

    # fusion flags copied onto the model config via LlmMetaConfig
    use_flash_attention: bool = True
    use_fused_rms_norm: bool = True
    use_fused_rope: bool = True
    use_fused_swiglu: bool = True
    use_fused_linear_cross_entropy: bool = False

    # dataloader
    dataloader_num_workers: int = 0
    dataloader_drop_last: bool = True
    distributed_dataloader: bool = False

    # checkpoint
    unified_checkpoint: bool = True
    # optimizer shards written by a shared-memory writer process
    # (reference unified_checkpoint async_save, :159-299)
    async_save: bool = False
    resume_from_checkpoint: Optional[str] = None
    save_on_each_node: bool = False
    ignore_data_skip: bool = False

    report_to: Optional[List[str]] = None
    run_name: Optional[str] = None
    disable_tqdm: bool = False

    # misc
    max_evaluate_steps: int = -1
    skip_profile_timer: bool = True

    # ---- reference-compatible config strings (training_args.py:655-690):
    # space-separated option switches parsed in __post_init__ and mapped
    # onto the native knobs
    tensor_parallel_config: str = ""
    pipeline_parallel_config: str = ""
    sharding_parallel_config: str = ""
    hybrid_parallel_topo_order: str = "dp_first"   # "dp_first" | "sharding_first"
    # fp16 initial loss scale (reference scale_loss)
    scale_loss: float = 2.0 ** 15
    fp16_opt_level: str = "O2"
    amp_custom_black_list: Optional[List[str]] = None
    amp_custom_white_list: Optional[List[str]] = None
    bf16_full_eval: bool = False
    eval_accumulation_steps: Optional[int] = None
    prediction_loss_only: bool = False
    metric_for_best_model: Optional[str] = None
    greater_is_better: Optional[bool] = None
    load_best_model_at_end: bool = False

    def __post_init__(self):
        self.world_size = int(os.environ.get("WORLD_SIZE", "1"))
        self.local_rank = int(os.environ.get("LOCAL_RANK", "0"))

        if self.fp16 and self.bf16:
            raise ValueError("Pick one of fp16/bf16")

        # reference-style option strings -> native switches
        self.tp_options = set(self.tensor_parallel_config.split())
        self.pp_options = set(self.pipeline_parallel_config.split())
        self.sd_options = set(self.sharding_parallel_config.split())
        known_tp = {"enable_mp_async_allreduce", "enable_mp_skip_c_identity",
                    "enable_mp_fused_linear_param_grad_add", "sync_param",
                    "sync_grad", "sync_moment"}
        known_pp = {"disable_p2p_cache_shape", "disable_partial_send_recv",
                    "enable_delay_scale_loss", "enable_dp_comm_overlap",
                    "enable_sharding_comm_overlap", "enable_release_grads",
                    "enable_timer", "enable_overlap_p2p_comm"}
        known_sd = {"enable_stage1_tensor_fusion", "enable_stage1_overlap",
                    "enable_stage2_overlap", "split_param",
                    "enable_stage1_broadcast_overlap", "disable_stage1_reduce_avg",
                    "enable_stage1_allgather_overlap"}
        for opts, known, name in ((self.tp_options, known_tp, "tensor"),
                                  (self.pp_options, known_pp, "pipeline"),
                                  (self.sd_options, known_sd, "sharding")):
            unknown = opts - known
            if unknown:
                logger.warning(f"ignoring unknown {name}_parallel_config "
                               f"options: {sorted(unknown)}")
        if {"enable_stage1_overlap", "enable_stage2_overlap",
            "enable_stage1_allgather_overlap"} & self.sd_options:
            self.sharding_overlap_comm = True
        if "enable_timer" in self.pp_options:
            self.skip_profile_timer = False
        # flat shard-aligned comm buffers are the native default, so
        # enable_stage1_tensor_fusion / split_param are already in effect

        # sharding options
        self.sharding_options = set()
        for opt in self.sharding.split():
            self.sharding_options.add(ShardingOption(opt))
        if self.sharding_parallel_degree == -1:
            self.sharding_parallel_degree = 1
        if self.sharding_options and self.sharding_parallel_degree == 1:
            # default: shard over everything left after tp/pp/sep
            denom = (
                self.tensor_parallel_degree
                * self.pipeline_parallel_degree
                * self.sep_parallel_degree
                * self.context_parallel_degree
            )
            self.sharding_parallel_degree = max(1, self.world_size // denom)
        if not self.sharding_options:
            self.sharding_parallel_degree = 1

        # sep and cp share one axis (reference training_args.py:1284-1286)
        if self.sep_parallel_degree > 1 and self.context_parallel_degree > 1:
            raise ValueError("sep_parallel_degree and context_parallel_degree share one axis")
        sep_axis = max(self.sep_parallel_degree, self.context_parallel_degree)

        denom = (
            self.tensor_parallel_degree
            * self.pipeline_parallel_degree
            * self.sharding_parallel_degree
            * sep_axis
        )
        if self.world_size % denom != 0:
            raise ValueError(
                f"world_size {self.world_size} not divisible by tp*pp*sharding*sep = {denom}"
            )
        self.data_parallel_degree = self.world_size // denom

        # build the topology (one process per GPU; RCCL on ROCm)
        self._topology = init_parallel_env(
            dp_degree=self.data_parallel_degree,
            pp_degree=self.pipeline_parallel_degree,
            sharding_degree=self.sharding_parallel_degree,
            sep_degree=sep_axis,
            mp_degree=self.tensor_parallel_degree,
        )

        if self.warmup_steps == 0 and self.warmup_ratio > 0 and self.max_steps > 0:
            self.warmup_steps = int(self.max_steps * self.warmup_ratio)

        if self.logging_dir is None:
            self.logging_dir = os.path.join(self.output_dir, "runs")

    # ------------------------------------------------------------------
    @property
    def topology(self) -> Topology:
        return getattr(self, "_topology", None) or get_topology()

    @property
    def device(self) -> torch.device:
        if torch.cuda.is_available():
            return torch.device("cuda", self.local_rank)
        return torch.device("cpu")

    @property
    def process_index(self) -> int:
        return self.topology.rank

    @property
    def dataset_world_size(self) -> int:
        return self.topology.dataset_world_size

    @property
    def dataset_rank(self) -> int:
        return self.topology.dataset_rank

    @property
    def train_batch_size(self) -> int:
        return self.per_device_train_batch_size

    @property
    def eval_batch_size(self) -> int:
        return self.per_device_eval_batch_size

    @property
    def global_train_batch_size(self) -> int:
        return (
            self.per_device_train_batch_size
            * self.gradient_accumulation_steps
            * self.dataset_world_size
        )

    @property
    def should_log(self) -> bool:
        return self.process_index == 0

    @property
    def should_save(self) -> bool:
        """Which ranks write model weights: mp rank 0, dp rank 0 by default;
        with unified checkpoint each tp rank writes its slice."""
        topo = self.topology
        if self.unified_checkpoint:
            return topo.coords.get("dp", 0) == 0
        return self.process_index == 0

    @property
    def compute_dtype(self) -> torch.dtype:
        if self.bf16:
            return torch.bfloat16
        if self.fp16:
            return torch.float16
        return torch.float32

    def sharding_stage(self) -> int:
        if ShardingOption.FULL_SHARD in self.sharding_options:
            return 3
        if ShardingOption.SHARD_GRAD_OP in self.sharding_options:
            return 2
        if ShardingOption.SHARD_OP in self.sharding_options:
            return 1
        return 0

    def to_dict(self):
        d = {}
        for k, v in self.__dict__.items():
            if k.startswith("_"):
                continue
            if isinstance(v, (set,)):
                v = sorted(str(x) for x in v)
            d[k] = v
        return d

    def to_json_string(self):
        return json.dumps(self.to_dict(), indent=2, default=str)

    def print_config(self):
        logger.info("=" * 60)
        logger.info("TrainingArguments")
        for k, v in sorted(self.to_dict().items()):
            logger.info(f"  {k:40s} = {v}")
        logger.info("=" * 60)
