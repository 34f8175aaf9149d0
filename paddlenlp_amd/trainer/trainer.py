"""Trainer: HF-style training loop with 4D-parallel wiring for MI355X.

Reference behavior: paddlenlp/trainer/trainer.py — Trainer.__init__ :273,
train :687, _inner_training_loop :855, training_step :2211, _wrap_model
:1895, _maybe_log_save_evaluate :1388, _save_checkpoint :2363,
evaluation_loop :2911.  The distributed runtime that the reference gets from
paddle fleet is implemented natively here: DP gradient sync via
parallel.data_parallel (bucketed RCCL all-reduce at the accumulation
boundary), ZeRO sharding via parallel.zero, bf16 "O2" via bf16 params +
fp32 master weights in FusedAdamW.
"""
from __future__ import annotations

import math
import os
import shutil
import time
from typing import Any, Callable, Dict, List, Optional, Tuple, Union

import torch
import torch.distributed as dist
import torch.nn as nn
from torch.utils.data import DataLoader, Dataset

from ..data import default_data_collator
from ..data.sampler import DistributedBatchSampler
from ..parallel.data_parallel import broadcast_parameters, fused_allreduce_gradients
from ..transformers.model_utils import PretrainedModel, unwrap_model
from ..utils.env import (
    OPTIMIZER_STATE_NAME,
    PREFIX_CHECKPOINT_DIR,
    SCHEDULER_NAME,
    TRAINER_STATE_NAME,
)
from ..utils.log import logger
from .optimizer import FusedAdamW
from .trainer_callback import (
    CallbackHandler,
    DefaultFlowCallback,
    ProgressCallback,
    TrainerControl,
    TrainerState,
)
from .integrations import get_reporting_integration_callbacks
from .plugins.timer import get_timers
from ..utils.profiler import add_profiler_step
from .trainer_utils import (
    DynamicLossScaler,
    TrainerMemoryTracker,
    TrainOutput,
    caculate_llm_flops,
    get_last_checkpoint,
    get_scheduler,
    set_hybrid_seed,
    should_skip_data,
    speed_metrics,
)
from .training_args import TrainingArguments

DEFAULT_CALLBACKS = [DefaultFlowCallback, ProgressCallback]


class Trainer:
    def __init__(
        self,
        model: nn.Module = None,
        args: TrainingArguments = None,
        data_collator: Optional[Callable] = None,
        train_dataset: Optional[Dataset] = None,
        eval_dataset: Optional[Dataset] = None,
        tokenizer=None,
        compute_metrics: Optional[Callable] = None,
        callbacks: Optional[List] = None,
        optimizers: Tuple = (None, None),
        criterion: Optional[nn.Module] = None,
    ):
        if args is None:
            args = TrainingArguments(output_dir="output")
        self.args = args
        self.model = model
        self.criterion = criterion
        self.data_collator = data_collator or default_data_collator
        self.train_dataset = train_dataset
        self.eval_dataset = eval_dataset
        self.tokenizer = tokenizer
        self.compute_metrics = compute_metrics
        self.optimizer, self.lr_scheduler = optimizers
        self.topology = args.topology

        set_hybrid_seed(args.seed, self.topology)

        self.state = TrainerState()
        self.state.is_world_process_zero = args.process_index == 0
        self.control = TrainerControl()
        callbacks = list(DEFAULT_CALLBACKS) + (callbacks or [])
        callbacks += get_reporting_integration_callbacks(args.report_to)
        self.callback_handler = CallbackHandler(
            callbacks, self.model, self.tokenizer, self.optimizer, self.lr_scheduler
        )
        self._model_wrapped = None
        self._zero = None  # ZeRO sharded-optimizer engine when sharding>0
        self.control = self.callback_handler.call_event(
            "on_init_end", args, self.state, self.control
        )

    # ------------------------------------------------------------------
    # dataloaders
    # ------------------------------------------------------------------
    def get_train_dataloader(self) -> DataLoader:
        sampler = DistributedBatchSampler(
            self.train_dataset,
            batch_size=self.args.per_device_train_batch_size,
            num_replicas=self.args.dataset_world_size,
            rank=self.args.dataset_rank,
            shuffle=False,
            drop_last=self.args.dataloader_drop_last,
            seed=self.args.seed,
            consumed_samples=self.state.consumed_samples,
        )
        if self.args.distributed_dataloader and (
            self.topology.mp_degree > 1 or self.topology.pp_degree > 1
            or self.topology.sep_degree > 1
        ):
            # only dp-source ranks read; mp/pp/sep peers receive broadcasts
            # (reference DistDataLoader, data/dist_dataloader.py:41)
            from ..data.dist_dataloader import DistDataLoader

            loader = DistDataLoader(
                self.train_dataset,
                batch_sampler=sampler,
                collate_fn=self.data_collator,
                num_workers=self.args.dataloader_num_workers,
                pin_memory=torch.cuda.is_available(),
                topology=self.topology,
            )
            loader.batch_sampler = sampler  # consumed-samples resume hook
            return loader
        return DataLoader(
            self.train_dataset,
            batch_sampler=sampler,
            collate_fn=self.data_collator,
            num_workers=self.args.dataloader_num_workers,
            pin_memory=torch.cuda.is_available(),
        )

    def get_eval_dataloader(self, eval_dataset=None) -> DataLoader:
        eval_dataset = eval_dataset or self.eval_dataset
        sampler = DistributedBatchSampler(
            eval_dataset,
            batch_size=self.args.per_device_eval_batch_size,
            num_replicas=self.args.dataset_world_size,
            rank=self.args.dataset_rank,
            shuffle=False,
            drop_last=False,
        )
        return DataLoader(
            eval_dataset,
            batch_sampler=sampler,
            collate_fn=self.data_collator,
            num_workers=self.args.dataloader_num_workers,
            pin_memory=torch.cuda.is_available(),
        )

    # ------------------------------------------------------------------
    # optimizer / scheduler
    # ------------------------------------------------------------------
    @staticmethod
    def _no_decay(name: str) -> bool:
        return name.endswith("bias") or "norm" in name.lower()

    def create_optimizer_and_scheduler(self, num_training_steps: int):
        if self.optimizer is None:
            model = unwrap_model(self.model)
            decay_params, no_decay_params = [], []
            for name, p in model.named_parameters():
                if not p.requires_grad:
                    continue
                p.param_name = name
                (no_decay_params if self._no_decay(name) else decay_params).append(p)
            groups = [
                {"params": decay_params, "weight_decay": self.args.weight_decay},
                {"params": no_decay_params, "weight_decay": 0.0},
            ]
            self.optimizer = FusedAdamW(
                groups,
                lr=self.args.learning_rate,
                betas=(self.args.adam_beta1, self.args.adam_beta2),
                eps=self.args.adam_epsilon,
                master_weights=self.args.bf16 or self.args.fp16,
            )
        if self.lr_scheduler is None:
            self.lr_scheduler = get_scheduler(
                self.args.lr_scheduler_type,
                self.optimizer,
                num_warmup_steps=self.args.warmup_steps,
                num_training_steps=num_training_steps,
                min_lr_ratio=self.args.min_lr_ratio,
            )
        self.callback_handler.optimizer = self.optimizer
        self.callback_handler.lr_scheduler = self.lr_scheduler

    # ------------------------------------------------------------------
    # model wrapping (reference _wrap_model :1895)
    # ------------------------------------------------------------------
    def _wrap_model(self, model: nn.Module) -> nn.Module:
        args = self.args
        device = args.device
        # bf16 "O2": cast parameters to bf16; FusedAdamW holds fp32 masters
        if args.bf16:
            model = model.to(dtype=torch.bfloat16)
        elif args.fp16:
            model = model.to(dtype=torch.float16)
        model = model.to(device)

        topo = self.topology
        if topo.pp_degree > 1:
            from ..parallel.pipeline import (
                InterleavedPipelineEngine,
                PipelineEngine,
                PipelineModule,
            )

            if not isinstance(unwrap_model(model), PipelineModule):
                raise ValueError(
                    "pipeline_parallel_degree > 1 requires a PipelineModule "
                    "(e.g. LlamaForCausalLMPipe)"
                )
            pipe = unwrap_model(model)
            hidden = pipe.config.hidden_size

            engine_cls = (InterleavedPipelineEngine
                          if pipe.num_virtual_stages > 1 else PipelineEngine)
            self._pipe_engine = engine_cls(
                pipe,
                hidden_shape_fn=lambda mb: (*mb["input_ids"].shape, hidden),
                dtype=args.compute_dtype,
                device=device,
            )
        if topo.dp_degree > 1 and topo.data_parallel_group is not None:
            broadcast_parameters(model, topo.data_parallel_group)
        if topo.sharding_degree > 1 and topo.sharding_parallel_group is not None:
            stage = args.sharding_stage()
            # engine is attached after optimizer creation in train()
            self._zero_stage = stage
            broadcast_parameters(model, topo.sharding_parallel_group)
        return model

    # ------------------------------------------------------------------
    # train
    # ------------------------------------------------------------------
    def train(self, resume_from_checkpoint: Optional[Union[str, bool]] = None):
        args = self.args
        self.is_in_train = True
        mem_tracker = TrainerMemoryTracker(skip=args.skip_memory_metrics)
        mem_tracker.start("train")
        self._loss_scaler = (DynamicLossScaler(init_scale=args.scale_loss)
                             if args.fp16 else None)

        train_dataloader = self.get_train_dataloader()
        steps_per_epoch = max(1, len(train_dataloader) // args.gradient_accumulation_steps)
        if args.max_steps > 0:
            max_steps = args.max_steps
            num_train_epochs = math.ceil(max_steps / steps_per_epoch)
        else:
            max_steps = int(steps_per_epoch * args.num_train_epochs)
            num_train_epochs = math.ceil(args.num_train_epochs)

        if resume_from_checkpoint is True or (
            resume_from_checkpoint is None and args.resume_from_checkpoint
        ):
            resume_from_checkpoint = args.resume_from_checkpoint or get_last_checkpoint(args.output_dir)
            # cross-rank completeness negotiation (reference all-reduces the
            # resume flag, trainer.py:711-721): with non-shared filesystems
            # ranks can see different checkpoints — every rank must resume
            # from the SAME one, or none at all
            if dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1:
                mine = resume_from_checkpoint if isinstance(resume_from_checkpoint, str) else None
                views = [None] * dist.get_world_size()
                dist.all_gather_object(views, mine)
                common = set(views)
                if len(common) > 1:
                    agreed = None
                    if None not in common:
                        # all ranks found something but disagree: take the
                        # oldest (the one every rank is guaranteed to have
                        # completed past)
                        def _step_of(p):
                            import re as _re

                            m = _re.search(r"(\d+)$", p.rstrip("/"))
                            return int(m.group(1)) if m else -1

                        agreed = min(common, key=_step_of)
                    logger.warning(
                        f"checkpoint negotiation: ranks saw {sorted(x or '<none>' for x in common)}; "
                        f"agreeing on {agreed or '<fresh start>'}")
                    resume_from_checkpoint = agreed

        model = self._wrap_model(self.model)
        self._model_wrapped = model
        self.callback_handler.model = model

        self.create_optimizer_and_scheduler(max_steps)

        # ZeRO engine wraps the optimizer after both exist
        if getattr(self, "_zero_stage", 0) in (1, 2):
            from ..parallel.zero import ZeroShardedEngine

            self._zero = ZeroShardedEngine(
                model, self.optimizer,
                stage=self._zero_stage,
                group=self.topology.sharding_parallel_group,
                bucket_mb=args.sharding_comm_buffer_size_MB,
            )
            if args.sharding_overlap_comm and self.topology.pp_degree == 1:
                self._zero.enable_overlap_comm()
        elif getattr(self, "_zero_stage", 0) == 3:
            from ..parallel.zero3 import Zero3Engine

            self._zero = Zero3Engine(
                unwrap_model(model), group=self.topology.sharding_parallel_group)

        if resume_from_checkpoint and isinstance(resume_from_checkpoint, str):
            self._load_from_checkpoint(resume_from_checkpoint)
            train_dataloader = self.get_train_dataloader()  # re-skip consumed samples

        model.train()

        total_batch_size = args.global_train_batch_size
        if args.should_log:
            logger.info("***** Running training *****")
            logger.info(f"  Num examples = {len(self.train_dataset):,}")
            logger.info(f"  Num epochs = {num_train_epochs}")
            logger.info(f"  Per-device batch size = {args.per_device_train_batch_size}")
            logger.info(f"  Global batch size (w. accumulation) = {total_batch_size}")
            logger.info(f"  Gradient accumulation steps = {args.gradient_accumulation_steps}")
            logger.info(f"  Total optimization steps = {max_steps:,}")

        self.state.max_steps = max_steps
        self.state.num_train_epochs = num_train_epochs
        self.callback_handler.train_dataloader = train_dataloader
        self.control = self.callback_handler.on_train_begin(args, self.state, self.control)

        tr_loss = torch.tensor(0.0, device=args.device)
        self._total_loss_scalar = 0.0
        self._globalstep_last_logged = self.state.global_step
        self._tokens_since_last_log = 0
        start_time = time.time()
        self._last_log_time = start_time

        epoch = self.state.epoch
        accum_count = 0
        pipe_buffer = []
        is_pipeline = self.topology.pp_degree > 1
        done = False
        while not done:
            for step, inputs in enumerate(train_dataloader):
                if accum_count == 0:
                    self.control = self.callback_handler.on_step_begin(args, self.state, self.control)

                # corrupted-data replay jump: consume the batch, train nothing
                if should_skip_data(self.state.global_step + 1,
                                    args.skip_data_intervals):
                    accum_count += 1
                    self.state.consumed_samples += (
                        args.per_device_train_batch_size * args.dataset_world_size)
                    if accum_count >= args.gradient_accumulation_steps:
                        accum_count = 0
                        self.state.global_step += 1
                        if self.state.global_step >= max_steps:
                            done = True
                            break
                    continue

                add_profiler_step()
                timers = get_timers()
                if is_pipeline:
                    # buffer micro-batches; the 1F1B engine consumes them as
                    # one optimizer step (reference training_pipeline_step
                    # trainer.py:2246-2290)
                    pipe_buffer.append(self._prepare_inputs(inputs))
                else:
                    timers("forward-backward").start()
                    if (self._zero is not None
                            and getattr(self._zero, "_overlap", False)
                            and accum_count == args.gradient_accumulation_steps - 1):
                        # final micro-batch: buckets reduce during backward
                        self._zero.overlap_active = True
                    loss = self.training_step(model, inputs)
                    timers("forward-backward").stop()
                    tr_loss += loss.detach()
                if "input_ids" in inputs:
                    self._tokens_since_last_log += inputs["input_ids"].numel()
                accum_count += 1
                self.state.consumed_samples += (
                    args.per_device_train_batch_size * args.dataset_world_size
                )

                if accum_count < args.gradient_accumulation_steps:
                    self.control = self.callback_handler.on_substep_end(args, self.state, self.control)
                    continue
                accum_count = 0

                if is_pipeline:
                    loss = self.training_pipeline_step(pipe_buffer)
                    pipe_buffer = []
                    tr_loss += loss.detach()

                self.optimizer_step(model)
                self.state.global_step += 1
                self.state.epoch = epoch + (step + 1) / max(1, len(train_dataloader))
                self.control = self.callback_handler.on_step_end(args, self.state, self.control)
                self._maybe_log_save_evaluate(tr_loss, model, start_time)

                if self.control.should_training_stop or self.state.global_step >= max_steps:
                    done = True
                    break
            epoch += 1
            # epoch end is a step boundary (reference semantics): flush a
            # partially filled gradient-accumulation window instead of
            # silently carrying grads into the next epoch / dropping the
            # final micro-batches
            if accum_count > 0 and not done:
                if is_pipeline and pipe_buffer:
                    loss = self.training_pipeline_step(pipe_buffer)
                    pipe_buffer = []
                    tr_loss += loss.detach()
                self.optimizer_step(model)
                self.state.global_step += 1
                accum_count = 0
                self.control = self.callback_handler.on_step_end(args, self.state, self.control)
                self._maybe_log_save_evaluate(tr_loss, model, start_time)
                if self.control.should_training_stop or self.state.global_step >= max_steps:
                    done = True
            # the consumed-samples resume skip applies only to the first
            # (resumed) epoch; later epochs start from sample 0
            sampler = getattr(train_dataloader, "batch_sampler", None)
            if sampler is not None and hasattr(sampler, "consumed_samples"):
                sampler.consumed_samples = 0
                sampler.set_epoch(int(epoch))
            self.control = self.callback_handler.on_epoch_end(args, self.state, self.control)
            # epoch-strategy eval/save fire from on_epoch_end's control
            # flags (reference runs the same hook after each epoch)
            self._maybe_log_save_evaluate(tr_loss, model, start_time)
            if epoch >= num_train_epochs:
                done = True

        if (args.load_best_model_at_end and self.state.best_model_checkpoint
                and os.path.isdir(self.state.best_model_checkpoint)):
            self._load_best_model()

        self.control = self.callback_handler.on_train_end(args, self.state, self.control)
        metrics = speed_metrics(
            "train", start_time,
            num_samples=self.state.consumed_samples,
            num_steps=self.state.global_step,
        )
        mem_tracker.stop_and_update_metrics(metrics)
        train_loss = self._total_loss_scalar / max(1, self.state.global_step)
        self.is_in_train = False
        return TrainOutput(self.state.global_step, train_loss, metrics)

    # ------------------------------------------------------------------
    def _prepare_inputs(self, inputs: Dict[str, Any]) -> Dict[str, Any]:
        device = self.args.device
        return {
            k: v.to(device, non_blocking=True) if isinstance(v, torch.Tensor) else v
            for k, v in inputs.items()
        }

    def compute_loss(self, model, inputs, return_outputs=False):
        """Reference trainer.py:2157."""
        labels = None
        if self.criterion is not None and "labels" in inputs:
            labels = inputs.pop("labels")
        outputs = model(**inputs)
        if self.criterion is not None and labels is not None:
            logits = outputs[0] if isinstance(outputs, tuple) else outputs
            loss = self.criterion(logits, labels)
            outputs = (loss, logits)
        elif isinstance(outputs, tuple):
            loss = outputs[0]
        elif isinstance(outputs, dict):
            loss = outputs["loss"]
        else:
            loss = outputs
        return (loss, outputs) if return_outputs else loss

    def training_pipeline_step(self, micro_batches) -> torch.Tensor:
        """One 1F1B pipeline pass over the buffered micro-batches."""
        loss = self._pipe_engine.forward_backward(
            micro_batches, input_fn=lambda mb: mb["input_ids"]
        )
        return loss

    def training_step(self, model: nn.Module, inputs: Dict[str, Any]) -> torch.Tensor:
        """Forward + backward for one micro-batch (reference :2211)."""
        model.train()
        inputs = self._prepare_inputs(inputs)
        if self.topology.sep_degree > 1:
            # sep/cp: every sep rank sees the same batch, sliced on seq
            # (reference split_inputs_sequence_dim, trainer.py:972-975)
            from ..parallel.segment_parallel import split_inputs_sequence_dim

            inputs = split_inputs_sequence_dim(
                inputs, self.topology.sep_parallel_group,
                # zigzag only applies to ring/context parallel; Ulysses
                # all-to-all needs contiguous chunks
                balanced=(self.args.context_parallel_balanced
                          and self.args.context_parallel_degree > 1))
        loss = self.compute_loss(model, inputs)
        if self.args.gradient_accumulation_steps > 1:
            loss = loss / self.args.gradient_accumulation_steps
        if getattr(self, "_loss_scaler", None) is not None:
            self._loss_scaler.scale_loss(loss).backward()
        else:
            loss.backward()
        return loss.detach()

    def optimizer_step(self, model: nn.Module):
        args = self.args
        topo = self.topology
        timers = get_timers()
        timers("all-reduce").start()
        # DP gradient sync (reference fused_allreduce_gradients :1079-1110)
        if self._zero is not None:
            self._zero.reduce_gradients_and_step_pre()
        if topo.dp_degree > 1:
            fused_allreduce_gradients(model.parameters(), topo.data_parallel_group)
        if topo.sep_degree > 1:
            # sep/cp ranks each backprop their seq slice of the same batch
            fused_allreduce_gradients(model.parameters(), topo.sep_parallel_group)
        if topo.mp_degree > 1 and getattr(self.model, "config", None) is not None and \
                getattr(unwrap_model(self.model).config, "sequence_parallel", False):
            # norm/bias params under sequence parallel see sharded activations
            for p in model.parameters():
                if getattr(p, "sequence_parallel", False) and p.grad is not None:
                    dist.all_reduce(p.grad, group=topo.model_parallel_group)

        timers("all-reduce").stop()
        if getattr(self, "_loss_scaler", None) is not None:
            finite = self._loss_scaler.unscale_and_check(
                [p for p in model.parameters() if p.grad is not None])
            # Under TP/PP/sharding ranks hold different gradient shards, so
            # the overflow verdict must be agreed globally or ranks diverge
            # into mismatched collectives (reference AMP all-reduces
            # found_inf across the hybrid groups).
            if dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1:
                flag = torch.tensor(
                    0.0 if finite else 1.0,
                    device=next(model.parameters()).device)
                dist.all_reduce(flag, op=dist.ReduceOp.MAX)
                finite = flag.item() == 0.0
            self._loss_scaler.update(found_inf=not finite)
            if not finite:
                # overflow: skip this step entirely (reference AMP behavior;
                # the LR schedule only advances when the optimizer ran)
                logger.warning(
                    f"fp16 overflow: skipping step, loss scale -> "
                    f"{self._loss_scaler.scale:g}")
                self._zero_grads()
                return
        if args.max_grad_norm and args.max_grad_norm > 0:
            if self._zero is not None and self._zero.stage == 3:
                self._zero.clip_grads(args.max_grad_norm)
            else:
                self._clip_grad_norm(model)

        timers("optimizer-step").start()
        self.optimizer.step()
        if self._zero is not None:
            self._zero.step_post()
        timers("optimizer-step").stop()
        self.lr_scheduler.step()
        self._zero_grads()

    def _zero_grads(self):
        if self._zero is not None and hasattr(self._zero, "zero_grad"):
            # flat-bucket engine: zero the persistent buffers and re-attach
            # grad views so backward accumulates in place (zero-copy)
            self._zero.zero_grad()
        else:
            self.optimizer.zero_grad(set_to_none=True)

    def _clip_grad_norm(self, model):
        """Global grad-norm clip, TP/PP-aware: local sum-of-squares, reduced
        over mp (sharded params) before the norm."""
        args = self.args
        topo = self.topology
        params = [p for p in model.parameters() if p.grad is not None]
        if not params:
            return
        device = params[0].grad.device
        local_sq = torch.zeros((), dtype=torch.float32, device=device)
        replicated_sq = torch.zeros((), dtype=torch.float32, device=device)
        for p in params:
            sq = p.grad.float().pow(2).sum()
            if getattr(p, "is_column_parallel", False) or getattr(p, "is_row_parallel", False):
                local_sq += sq
            else:
                replicated_sq += sq
        total = local_sq.clone()
        if topo.mp_degree > 1 and topo.model_parallel_group is not None:
            dist.all_reduce(total, group=topo.model_parallel_group)
        total += replicated_sq
        if topo.pp_degree > 1 and topo.pipe_parallel_group is not None:
            dist.all_reduce(total, group=topo.pipe_parallel_group)
        if self._zero is not None and self._zero.stage >= 1:
            # grads are reduce-scattered for stage 1 and 2 alike: each rank
            # holds only its owned shard at clip time
            dist.all_reduce(total, group=topo.sharding_parallel_group)
        norm = total.sqrt()
        clip = (args.max_grad_norm / (norm + 1e-6)).clamp(max=1.0)
        for p in params:
            p.grad.mul_(clip.to(p.grad.dtype))

    # ------------------------------------------------------------------
    # logging / save / evaluate
    # ------------------------------------------------------------------
    def _model_flops_per_step(self) -> Optional[float]:
        model = unwrap_model(self.model)
        cfg = getattr(model, "config", None)
        if cfg is None or not hasattr(cfg, "num_hidden_layers"):
            return None
        seq_len = getattr(cfg, "max_position_embeddings", 4096)
        seq_len = getattr(self, "_observed_seq_len", seq_len)
        tokens = self._tokens_since_last_log
        if tokens == 0:
            return None
        batch_equiv = tokens / seq_len
        return caculate_llm_flops(
            cfg.hidden_size, cfg.intermediate_size, cfg.num_hidden_layers,
            cfg.vocab_size, seq_len, batch_size=batch_equiv,
            recompute=getattr(cfg, "recompute", False),
        )

    def _maybe_log_save_evaluate(self, tr_loss, model, start_time):
        args, state, control = self.args, self.state, self.control
        if control.should_log:
            steps = state.global_step - self._globalstep_last_logged
            loss_val = tr_loss.item() / max(1, steps) / args.gradient_accumulation_steps * args.gradient_accumulation_steps
            loss_scalar = (tr_loss / max(1, steps)).item()
            tr_loss.zero_()
            self._total_loss_scalar += loss_scalar * steps
            interval = time.time() - self._last_log_time
            logs = {
                "loss": round(loss_scalar, 6),
                "learning_rate": self.lr_scheduler.get_last_lr()[0],
                "global_step": state.global_step,
            }
            if interval > 0 and self._tokens_since_last_log:
                logs["interval_tokens_per_second_per_device"] = round(
                    self._tokens_since_last_log / interval, 1
                )
                flops = self._model_flops_per_step()
                if flops:
                    logs["interval_hardware_tflops_per_device"] = round(flops / interval / 1e12, 1)
            if torch.cuda.is_available():
                logs["gpu_mem_max_allocated_gb"] = round(
                    torch.cuda.max_memory_allocated() / 1e9, 2
                )
            self._tokens_since_last_log = 0
            self._last_log_time = time.time()
            self._globalstep_last_logged = state.global_step
            if not args.skip_profile_timer:
                timer_msg = get_timers().log()
                if timer_msg:
                    logs["timers"] = timer_msg
            self.log(logs)
        metrics = None
        if control.should_evaluate and self.eval_dataset is not None:
            metrics = self.evaluate()
            self.control = self.callback_handler.on_evaluate(args, state, control, metrics=metrics)
        if control.should_save:
            self._save_checkpoint(model)
            self._update_best_checkpoint(metrics)
            self.control = self.callback_handler.on_save(args, state, self.control)

    def _update_best_checkpoint(self, metrics):
        """Track best checkpoint for load_best_model_at_end (reference
        trainer.py:2464-2477)."""
        args = self.args
        if not args.metric_for_best_model or metrics is None:
            return
        key = args.metric_for_best_model
        if not key.startswith("eval_"):
            key = f"eval_{key}"
        if key not in metrics:
            return
        value = metrics[key]
        greater = args.greater_is_better
        if greater is None:
            greater = not key.endswith("loss")
        best = self.state.best_metric
        if best is None or (value > best if greater else value < best):
            self.state.best_metric = value
            self.state.best_model_checkpoint = self._checkpoint_dir()

    def log_metrics(self, split: str, metrics: Dict[str, float]):
        """Pretty-print a metrics dict (reference log_metrics helper)."""
        if not self.args.process_index == 0:
            return
        logger.info(f"***** {split} metrics *****")
        for k in sorted(metrics):
            logger.info(f"  {k} = {metrics[k]}")

    def save_metrics(self, split: str, metrics: Dict[str, float]):
        """Write metrics to <output_dir>/<split>_results.json and append to
        all_results.json (reference save_metrics helper)."""
        if not self.args.process_index == 0:
            return
        import json as _json

        path = os.path.join(self.args.output_dir, f"{split}_results.json")
        os.makedirs(self.args.output_dir, exist_ok=True)
        with open(path, "w") as f:
            _json.dump(metrics, f, indent=2, sort_keys=True)
        all_path = os.path.join(self.args.output_dir, "all_results.json")
        merged = {}
        if os.path.isfile(all_path):
            with open(all_path) as f:
                merged = _json.load(f)
        merged.update(metrics)
        with open(all_path, "w") as f:
            _json.dump(merged, f, indent=2, sort_keys=True)

    def save_state(self):
        """Persist trainer_state.json to output_dir (reference helper)."""
        if self.args.process_index == 0:
            os.makedirs(self.args.output_dir, exist_ok=True)
            self.state.save_to_json(
                os.path.join(self.args.output_dir, TRAINER_STATE_NAME))

    def log(self, logs: Dict[str, float]):
        logs["epoch"] = round(self.state.epoch, 4)
        self.state.log_history.append(dict(logs))
        self.control = self.callback_handler.on_log(self.args, self.state, self.control, logs=logs)

    # ------------------------------------------------------------------
    # checkpointing (reference _save_checkpoint :2363)
    # ------------------------------------------------------------------
    def _checkpoint_dir(self) -> str:
        return os.path.join(self.args.output_dir, f"{PREFIX_CHECKPOINT_DIR}-{self.state.global_step}")

    def _save_checkpoint(self, model):
        args = self.args
        ckpt_dir = self._checkpoint_dir()
        os.makedirs(ckpt_dir, exist_ok=True)
        self.save_model(ckpt_dir)

        if args.unified_checkpoint:
            from .unified_checkpoint import save_unified_optimizer

            saver = None
            if args.async_save:
                if not hasattr(self, "_shm_saver"):
                    from .utils.shm_save import ShmAsyncSaver

                    self._shm_saver = ShmAsyncSaver()
                saver = self._shm_saver
            save_unified_optimizer(self.optimizer, unwrap_model(model), ckpt_dir,
                                   self.topology, zero=self._zero, saver=saver)
        elif args.process_index == 0 or (self._zero is not None):
            torch.save(self.optimizer.state_dict(), os.path.join(ckpt_dir, OPTIMIZER_STATE_NAME))

        if args.process_index == 0:
            torch.save(self.lr_scheduler.state_dict(), os.path.join(ckpt_dir, SCHEDULER_NAME))
            if getattr(self, "_loss_scaler", None) is not None:
                torch.save(self._loss_scaler.state_dict(),
                           os.path.join(ckpt_dir, "scaler.pt"))
            self.state.save_to_json(os.path.join(ckpt_dir, TRAINER_STATE_NAME))
            self._save_rng_state(ckpt_dir)
            self._rotate_checkpoints()
        if getattr(self, "_shm_saver", None) is not None:
            # the marker below asserts completeness: drain the writer first
            self._shm_saver.wait_all()
        if dist.is_initialized():
            dist.barrier()
        # integrity marker AFTER every rank finished writing (reference PDC
        # .checkpoint_done, trainer.py:2520-2524); get_last_checkpoint only
        # resumes from marked checkpoints
        if args.process_index == 0:
            from .trainer_utils import CHECKPOINT_DONE_MARKER

            with open(os.path.join(ckpt_dir, CHECKPOINT_DONE_MARKER), "w") as f:
                f.write(str(self.state.global_step))

    def _save_rng_state(self, ckpt_dir):
        rng = {
            "python": __import__("random").getstate(),
            "numpy": __import__("numpy").random.get_state(),
            "torch": torch.get_rng_state(),
        }
        if torch.cuda.is_available():
            rng["cuda"] = torch.cuda.get_rng_state_all()
        torch.save(rng, os.path.join(ckpt_dir, f"rng_state_{self.topology.world_size}.pth"))

    def _rotate_checkpoints(self):
        limit = self.args.save_total_limit
        if not limit:
            return
        import re

        pat = re.compile(rf"^{PREFIX_CHECKPOINT_DIR}-(\d+)$")
        ckpts = sorted(
            (
                d for d in os.listdir(self.args.output_dir)
                if pat.match(d) and os.path.isdir(os.path.join(self.args.output_dir, d))
            ),
            key=lambda d: int(pat.match(d).group(1)),
        )
        # never rotate away the best checkpoint (load_best_model_at_end)
        best = self.state.best_model_checkpoint
        if best is not None:
            best = os.path.basename(os.path.normpath(best))
            if best in ckpts:
                ckpts.remove(best)
        for d in ckpts[:-limit]:
            shutil.rmtree(os.path.join(self.args.output_dir, d), ignore_errors=True)

    def save_model(self, output_dir: Optional[str] = None):
        output_dir = output_dir or self.args.output_dir
        model = unwrap_model(self.model)
        if self.args.unified_checkpoint:
            from .unified_checkpoint import save_unified_model

            save_unified_model(model, output_dir, self.topology)
        elif self.args.process_index == 0:
            if isinstance(model, PretrainedModel):
                model.save_pretrained(output_dir)
            else:
                torch.save(model.state_dict(), os.path.join(output_dir, "pytorch_model.bin"))
        if self.tokenizer is not None and self.args.process_index == 0:
            self.tokenizer.save_pretrained(output_dir)

    def _load_best_model(self):
        """Reload the best checkpoint's model weights at train end
        (reference _load_best_model, trainer.py:2490-2516)."""
        ckpt_dir = self.state.best_model_checkpoint
        logger.info(
            f"Loading best model from {ckpt_dir} "
            f"({self.args.metric_for_best_model}={self.state.best_metric})")
        model = unwrap_model(self.model)
        if self.args.unified_checkpoint:
            from .unified_checkpoint import load_unified_checkpoint

            load_unified_checkpoint(model, None, ckpt_dir, self.topology)
        else:
            from safetensors.torch import load_file

            path = os.path.join(ckpt_dir, "model.safetensors")
            if os.path.isfile(path):
                model.load_state_dict(load_file(path), strict=False)
            else:
                path = os.path.join(ckpt_dir, "pytorch_model.bin")
                model.load_state_dict(
                    torch.load(path, weights_only=True), strict=False)

    def _load_from_checkpoint(self, ckpt_dir: str):
        logger.info(f"Resuming from checkpoint {ckpt_dir}")
        model = unwrap_model(self.model)
        if self.args.unified_checkpoint:
            from .unified_checkpoint import load_unified_checkpoint

            load_unified_checkpoint(model, self.optimizer, ckpt_dir, self.topology, zero=self._zero)
        else:
            from safetensors.torch import load_file

            state = load_file(os.path.join(ckpt_dir, "model.safetensors"))
            model.load_state_dict(state, strict=False)
            opt_path = os.path.join(ckpt_dir, OPTIMIZER_STATE_NAME)
            if os.path.isfile(opt_path):
                self.optimizer.load_state_dict(torch.load(opt_path, weights_only=False))
        sched_path = os.path.join(ckpt_dir, SCHEDULER_NAME)
        if os.path.isfile(sched_path):
            self.lr_scheduler.load_state_dict(torch.load(sched_path, weights_only=False))
            # re-apply the restored LR now: the optimizer keeps its
            # construction-time LR until the next scheduler.step() otherwise
            for group, lr in zip(self.optimizer.param_groups, self.lr_scheduler.get_last_lr()):
                group["lr"] = lr
        scaler_path = os.path.join(ckpt_dir, "scaler.pt")
        if os.path.isfile(scaler_path) and getattr(self, "_loss_scaler", None) is not None:
            self._loss_scaler.load_state_dict(
                torch.load(scaler_path, weights_only=False))
        state_path = os.path.join(ckpt_dir, TRAINER_STATE_NAME)
        if os.path.isfile(state_path):
            self.state = TrainerState.load_from_json(state_path)
            self.state.is_world_process_zero = self.args.process_index == 0
        rng_path = os.path.join(ckpt_dir, f"rng_state_{self.topology.world_size}.pth")
        if os.path.isfile(rng_path):
            rng = torch.load(rng_path, weights_only=False)
            __import__("random").setstate(rng["python"])
            __import__("numpy").random.set_state(rng["numpy"])
            torch.set_rng_state(rng["torch"])
            if torch.cuda.is_available() and "cuda" in rng:
                try:
                    torch.cuda.set_rng_state_all(rng["cuda"])
                except RuntimeError:
                    logger.warning("CUDA rng restore skipped (device count changed)")

    # ------------------------------------------------------------------
    # evaluation
    # ------------------------------------------------------------------
    def evaluate(self, eval_dataset=None, metric_key_prefix: str = "eval") -> Dict[str, float]:
        model = self._model_wrapped or self.model
        dataloader = self.get_eval_dataloader(eval_dataset)
        model.eval()
        losses = []
        collect = self.compute_metrics is not None
        all_logits, all_labels = [], []
        start = time.time()
        n_samples = 0
        with torch.no_grad():
            for i, inputs in enumerate(dataloader):
                if 0 < self.args.max_evaluate_steps <= i:
                    break
                inputs = self._prepare_inputs(inputs)
                if collect:
                    loss, outputs = self.compute_loss(model, inputs,
                                                      return_outputs=True)
                    logits = outputs[1] if isinstance(outputs, tuple) else outputs
                    if isinstance(logits, torch.Tensor):
                        all_logits.append(logits.detach().float().cpu())
                    if "labels" in inputs:
                        all_labels.append(inputs["labels"].detach().cpu())
                else:
                    loss = self.compute_loss(model, inputs)
                losses.append(loss.float())
                n_samples += next(iter(inputs.values())).shape[0]
        model.train()
        mean_loss = torch.stack(losses).mean() if losses else torch.tensor(float("nan"))
        if dist.is_initialized() and self.topology.data_parallel_group is not None:
            dist.all_reduce(mean_loss, group=self.topology.data_parallel_group)
            mean_loss /= dist.get_world_size(self.topology.data_parallel_group)
        metrics = {f"{metric_key_prefix}_loss": mean_loss.item()}
        if collect and all_logits:
            # reference evaluation_loop hands (predictions, labels) to
            # compute_metrics and prefixes the result
            extra = self.compute_metrics(
                (torch.cat(all_logits),
                 torch.cat(all_labels) if all_labels else None))
            for k, v in (extra or {}).items():
                key = k if k.startswith(metric_key_prefix) else                     f"{metric_key_prefix}_{k}"
                metrics[key] = v
        try:
            metrics[f"{metric_key_prefix}_ppl"] = math.exp(mean_loss.item())
        except OverflowError:
            metrics[f"{metric_key_prefix}_ppl"] = float("inf")
        metrics.update(speed_metrics(metric_key_prefix, start, num_samples=n_samples))
        self.log(metrics)
        return metrics

    def predict(self, test_dataset, metric_key_prefix: str = "test"):
        model = self._model_wrapped or self.model
        dataloader = self.get_eval_dataloader(test_dataset)
        model.eval()
        all_logits, all_labels = [], []
        with torch.no_grad():
            for inputs in dataloader:
                inputs = self._prepare_inputs(inputs)
                labels = inputs.get("labels")
                loss, outputs = self.compute_loss(model, inputs, return_outputs=True)
                logits = outputs[1] if isinstance(outputs, tuple) and len(outputs) > 1 else outputs
                all_logits.append(logits.float().cpu())
                if labels is not None:
                    all_labels.append(labels.cpu())
        model.train()
        logits = torch.cat(all_logits) if all_logits else None
        labels = torch.cat(all_labels) if all_labels else None
        # merge across data-parallel ranks (reference _nested_gather /
        # distributed_concat, trainer.py:3302): each dp rank saw a disjoint
        # dataset shard; every rank returns the full concatenation
        dp_group = self.topology.data_parallel_group
        if dist.is_initialized() and dp_group is not None and \
                dist.get_world_size(dp_group) > 1:
            gathered = [None] * dist.get_world_size(dp_group)
            dist.all_gather_object(gathered, (logits, labels), group=dp_group)
            lg = [g[0] for g in gathered if g[0] is not None]
            lb = [g[1] for g in gathered if g[1] is not None]
            logits = torch.cat(lg) if lg else logits
            labels = torch.cat(lb) if lb else labels
        metrics = {}
        if self.compute_metrics is not None and logits is not None:
            metrics = self.compute_metrics((logits, labels))
        return logits, labels, metrics
