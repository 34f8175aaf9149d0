"""Pipeline-parallel tests (gloo, world 2): 1F1B loss/grad parity with the
single-process model."""
import os

import torch
import torch.distributed as dist

from tests.test_distributed import _run_workers


def _w_pp_llama(rank, world):
    from paddlenlp_amd.parallel.topology import init_parallel_env
    from paddlenlp_amd.parallel.pipeline import PipelineEngine
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM
    from paddlenlp_amd.transformers.llama.modeling_pp import LlamaForCausalLMPipe

    topo = init_parallel_env(pp_degree=world, backend="gloo")
    cfg = LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, dtype="float32",
    )
    torch.manual_seed(5)
    full = LlamaForCausalLM.from_config(cfg)
    base_sd = full.state_dict()

    pipe = LlamaForCausalLMPipe(cfg)
    pipe.load_base_state_dict(base_sd)
    engine = PipelineEngine(
        pipe,
        hidden_shape_fn=lambda mb: (*mb["input_ids"].shape, cfg.hidden_size),
        dtype=torch.float32,
        device=torch.device("cpu"),
    )

    g = torch.Generator().manual_seed(9)
    micro_batches = []
    for _ in range(4):
        ids = torch.randint(0, 128, (2, 16), generator=g)
        labels = torch.randint(0, 128, (2, 16), generator=g)
        micro_batches.append({"input_ids": ids, "labels": labels})

    loss = engine.forward_backward(micro_batches, input_fn=lambda mb: mb["input_ids"])

    # single-process reference
    ref_losses = []
    for mb in micro_batches:
        l, _ = full(input_ids=mb["input_ids"], labels=mb["labels"])
        (l / len(micro_batches)).backward()
        ref_losses.append(l.detach())
    ref_loss = torch.stack(ref_losses).mean()

    if pipe.is_last_stage:
        assert torch.allclose(loss, ref_loss, atol=1e-5), (loss, ref_loss)

    # gradient parity for this stage's params
    name_map = pipe.pp_param_name_map()
    ref_params = dict(full.named_parameters())
    checked = 0
    for local_name, p in pipe.named_parameters():
        base_name = name_map[local_name]
        ref_g = ref_params[base_name].grad
        assert p.grad is not None, local_name
        assert torch.allclose(p.grad, ref_g, atol=1e-5), (
            base_name, (p.grad - ref_g).abs().max())
        checked += 1
    assert checked > 0


def _w_pp_uneven(rank, world):
    """M < P warmup edge + M not divisible by anything."""
    from paddlenlp_amd.parallel.topology import init_parallel_env
    from paddlenlp_amd.parallel.pipeline import PipelineEngine
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM
    from paddlenlp_amd.transformers.llama.modeling_pp import LlamaForCausalLMPipe

    topo = init_parallel_env(pp_degree=world, backend="gloo")
    cfg = LlamaConfig(
        vocab_size=64, hidden_size=32, intermediate_size=64,
        num_hidden_layers=2, num_attention_heads=2, num_key_value_heads=2,
        max_position_embeddings=32, dtype="float32",
    )
    torch.manual_seed(5)
    full = LlamaForCausalLM.from_config(cfg)
    pipe = LlamaForCausalLMPipe(cfg)
    pipe.load_base_state_dict(full.state_dict())
    engine = PipelineEngine(
        pipe, hidden_shape_fn=lambda mb: (*mb["input_ids"].shape, cfg.hidden_size),
        dtype=torch.float32, device=torch.device("cpu"),
    )
    g = torch.Generator().manual_seed(2)
    mbs = [{"input_ids": torch.randint(0, 64, (1, 8), generator=g),
            "labels": torch.randint(0, 64, (1, 8), generator=g)}]  # M=1 < P=2
    loss = engine.forward_backward(mbs, input_fn=lambda mb: mb["input_ids"])
    if pipe.is_last_stage:
        ref, _ = full(input_ids=mbs[0]["input_ids"], labels=mbs[0]["labels"])
        assert torch.allclose(loss, ref.detach(), atol=1e-5)


def test_pp_llama_1f1b_parity():
    _run_workers(_w_pp_llama)


def test_pp_single_microbatch():
    _run_workers(_w_pp_uneven)


def _w_tp2_pp2(rank, world):
    """TP2 x PP2 (world 4): pipe forward/backward parity with single-process."""
    import os, shutil
    from paddlenlp_amd.parallel.topology import init_parallel_env
    from paddlenlp_amd.parallel.pipeline import PipelineEngine
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM
    from paddlenlp_amd.transformers.llama.modeling_pp import LlamaForCausalLMPipe

    topo = init_parallel_env(pp_degree=2, mp_degree=2, backend="gloo")
    base_cfg = dict(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, dtype="float32",
    )
    torch.manual_seed(5)
    full = LlamaForCausalLM.from_config(LlamaConfig(**base_cfg))
    tmp = "/tmp/pdnlp_tp_pp_test"
    if rank == 0:
        os.makedirs(tmp, exist_ok=True)
        full.save_pretrained(tmp)
    dist.barrier()

    cfg = LlamaConfig(**{**base_cfg, "tensor_parallel_degree": 2,
                         "tensor_parallel_rank": topo.get_rank_in("mp"),
                         "tensor_parallel_output": False})
    pipe = LlamaForCausalLMPipe(cfg)
    # load TP-split weights through the tp mappings
    from safetensors.torch import load_file
    base_sd = load_file(os.path.join(tmp, "model.safetensors"))
    actions = LlamaForCausalLM._get_tensor_parallel_mappings(cfg, is_split=True)
    split_sd = {k: (actions[k](v) if k in actions else v) for k, v in base_sd.items()}
    pipe.load_base_state_dict(split_sd)

    engine = PipelineEngine(
        pipe, hidden_shape_fn=lambda mb: (*mb["input_ids"].shape, cfg.hidden_size),
        dtype=torch.float32, device=torch.device("cpu"),
    )
    g = torch.Generator().manual_seed(9)
    mbs = []
    for _ in range(2):
        mbs.append({"input_ids": torch.randint(0, 128, (2, 16), generator=g),
                    "labels": torch.randint(0, 128, (2, 16), generator=g)})
    loss = engine.forward_backward(mbs, input_fn=lambda mb: mb["input_ids"])

    ref_losses = []
    for mb in mbs:
        l, _ = full(input_ids=mb["input_ids"], labels=mb["labels"])
        ref_losses.append(l.detach())
    ref = torch.stack(ref_losses).mean()
    if pipe.is_last_stage:
        assert torch.allclose(loss, ref, atol=1e-4), (loss, ref)
    dist.barrier()
    if rank == 0:
        shutil.rmtree(tmp, ignore_errors=True)


def test_tp2_pp2_parity():
    _run_workers(_w_tp2_pp2, world_size=4)


def _w_vpp_llama(rank, world, v=2, n_mb=4):
    """Interleaved (VPP) schedule: loss + grad parity with single process."""
    from paddlenlp_amd.parallel.topology import init_parallel_env
    from paddlenlp_amd.parallel.pipeline import InterleavedPipelineEngine
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM
    from paddlenlp_amd.transformers.llama.modeling_pp import LlamaForCausalLMPipe

    topo = init_parallel_env(pp_degree=world, backend="gloo")
    cfg = LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=8, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, dtype="float32",
    )
    torch.manual_seed(5)
    full = LlamaForCausalLM.from_config(cfg)
    base_sd = full.state_dict()

    pipe = LlamaForCausalLMPipe(cfg, num_virtual_stages=v)
    assert len(pipe.chunk_bounds) == v
    pipe.load_base_state_dict(base_sd)
    engine = InterleavedPipelineEngine(
        pipe,
        hidden_shape_fn=lambda mb: (*mb["input_ids"].shape, cfg.hidden_size),
        dtype=torch.float32,
        device=torch.device("cpu"),
    )

    g = torch.Generator().manual_seed(9)
    micro_batches = []
    for _ in range(n_mb):
        ids = torch.randint(0, 128, (2, 16), generator=g)
        labels = torch.randint(0, 128, (2, 16), generator=g)
        micro_batches.append({"input_ids": ids, "labels": labels})

    loss = engine.forward_backward(micro_batches, input_fn=lambda mb: mb["input_ids"])

    ref_losses = []
    for mb in micro_batches:
        l, _ = full(input_ids=mb["input_ids"], labels=mb["labels"])
        (l / len(micro_batches)).backward()
        ref_losses.append(l.detach())
    ref_loss = torch.stack(ref_losses).mean()

    # last virtual stage lives on rank (V-1) % P = world-1
    if rank == world - 1:
        assert torch.allclose(loss, ref_loss, atol=1e-5), (loss, ref_loss)

    name_map = pipe.pp_param_name_map()
    ref_params = dict(full.named_parameters())
    checked = 0
    for local_name, p in pipe.named_parameters():
        ref_g = ref_params[name_map[local_name]].grad
        assert p.grad is not None, local_name
        assert torch.allclose(p.grad, ref_g, atol=1e-5), (
            name_map[local_name], (p.grad - ref_g).abs().max())
        checked += 1
    assert checked > 0


def test_vpp_interleaved_parity():
    _run_workers(_w_vpp_llama)


def test_vpp_v3_min_microbatches():
    _run_workers(_w_vpp_llama, extra=(3, 2))


def _w_qwen2_pipe(rank, world):
    """Qwen2 pipe variant: world-2 1F1B parity with the single-process model."""
    from paddlenlp_amd.parallel.topology import init_parallel_env
    from paddlenlp_amd.parallel.pipeline import PipelineEngine
    from paddlenlp_amd.transformers import Qwen2Config, Qwen2ForCausalLM
    from paddlenlp_amd.transformers.llama.modeling_pp import Qwen2ForCausalLMPipe

    topo = init_parallel_env(pp_degree=world, backend="gloo")
    cfg = Qwen2Config(
        vocab_size=96, hidden_size=32, intermediate_size=64,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, dtype="float32")
    torch.manual_seed(5)
    full = Qwen2ForCausalLM.from_config(cfg)
    pipe = Qwen2ForCausalLMPipe(cfg)
    pipe.load_base_state_dict(full.state_dict())
    engine = PipelineEngine(
        pipe, hidden_shape_fn=lambda mb: (*mb["input_ids"].shape, cfg.hidden_size),
        dtype=torch.float32, device=torch.device("cpu"))

    g = torch.Generator().manual_seed(9)
    mbs = [{"input_ids": torch.randint(0, 96, (2, 8), generator=g),
            "labels": torch.randint(0, 96, (2, 8), generator=g)}
           for _ in range(2)]
    loss = engine.forward_backward(mbs, input_fn=lambda mb: mb["input_ids"])
    ref_losses = []
    for mb in mbs:
        l, _ = full(input_ids=mb["input_ids"], labels=mb["labels"])
        (l / len(mbs)).backward()
        ref_losses.append(l.detach())
    if pipe.is_last_stage:
        assert torch.allclose(loss, torch.stack(ref_losses).mean(), atol=1e-5)
    name_map = pipe.pp_param_name_map()
    ref_params = dict(full.named_parameters())
    for local_name, p in pipe.named_parameters():
        assert torch.allclose(p.grad, ref_params[name_map[local_name]].grad,
                              atol=1e-5), local_name


def test_qwen2_pipe_parity():
    _run_workers(_w_qwen2_pipe)


def _w_trainer_vpp(rank, world):
    """Trainer glue: virtual_pp_degree>1 selects the interleaved engine and
    trains through it."""
    from paddlenlp_amd.parallel.pipeline import InterleavedPipelineEngine
    from paddlenlp_amd.parallel.topology import init_parallel_env
    from paddlenlp_amd.trainer import Trainer
    from paddlenlp_amd.trainer.training_args import TrainingArguments
    from paddlenlp_amd.transformers import LlamaConfig
    from paddlenlp_amd.transformers.llama.modeling_pp import LlamaForCausalLMPipe

    init_parallel_env(pp_degree=world, backend="gloo")
    cfg = LlamaConfig(
        vocab_size=96, hidden_size=32, intermediate_size=64,
        num_hidden_layers=8, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, dtype="float32")
    torch.manual_seed(0)
    pipe = LlamaForCausalLMPipe(cfg, num_virtual_stages=2)
    data = [{"input_ids": torch.randint(0, 96, (8,)),
             "labels": torch.randint(0, 96, (8,))} for _ in range(8)]
    import tempfile

    with tempfile.TemporaryDirectory() as out:
        args = TrainingArguments(
            output_dir=out, do_train=True, max_steps=2,
            per_device_train_batch_size=1, gradient_accumulation_steps=2,
            pipeline_parallel_degree=world, virtual_pp_degree=2,
            logging_steps=100, save_steps=1 << 30, report_to=[])
        trainer = Trainer(model=pipe, args=args, train_dataset=data)
        trainer.train()
        assert isinstance(trainer._pipe_engine, InterleavedPipelineEngine)
        assert trainer.state.global_step == 2
        # every local param received gradients through the vpp schedule
        # (grads are zeroed post-step; weights must have moved instead)


def test_trainer_vpp_glue():
    _run_workers(_w_trainer_vpp)


def _w_tp2_pp2_sp(rank, world):
    """TP2 x PP2 + sequence parallel (world 4): the SP-sharded [B, s/mp, H]
    activation crosses the stage boundary via the shape-negotiated p2p."""
    import os, shutil
    from paddlenlp_amd.parallel.topology import init_parallel_env
    from paddlenlp_amd.parallel.pipeline import PipelineEngine
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM
    from paddlenlp_amd.transformers.llama.modeling_pp import LlamaForCausalLMPipe

    topo = init_parallel_env(pp_degree=2, mp_degree=2, backend="gloo")
    base_cfg = dict(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=4, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, dtype="float32",
    )
    torch.manual_seed(5)
    full = LlamaForCausalLM.from_config(LlamaConfig(**base_cfg))
    tmp = "/tmp/pdnlp_tp_pp_sp_test"
    if rank == 0:
        os.makedirs(tmp, exist_ok=True)
        full.save_pretrained(tmp)
    dist.barrier()

    cfg = LlamaConfig(**{**base_cfg, "tensor_parallel_degree": 2,
                         "tensor_parallel_rank": topo.get_rank_in("mp"),
                         "tensor_parallel_output": False,
                         "sequence_parallel": True})
    pipe = LlamaForCausalLMPipe(cfg)
    from safetensors.torch import load_file
    base_sd = load_file(os.path.join(tmp, "model.safetensors"))
    actions = LlamaForCausalLM._get_tensor_parallel_mappings(cfg, is_split=True)
    split_sd = {k: (actions[k](v) if k in actions else v) for k, v in base_sd.items()}
    pipe.load_base_state_dict(split_sd)

    engine = PipelineEngine(
        pipe, hidden_shape_fn=None, dtype=torch.float32,
        device=torch.device("cpu"),
    )
    g = torch.Generator().manual_seed(9)
    mbs = []
    for _ in range(2):
        mbs.append({"input_ids": torch.randint(0, 128, (2, 16), generator=g),
                    "labels": torch.randint(0, 128, (2, 16), generator=g)})
    loss = engine.forward_backward(mbs, input_fn=lambda mb: mb["input_ids"])

    ref_losses = []
    for mb in mbs:
        l, _ = full(input_ids=mb["input_ids"], labels=mb["labels"])
        ref_losses.append(l.detach())
    ref = torch.stack(ref_losses).mean()
    if pipe.is_last_stage:
        assert torch.allclose(loss, ref, atol=1e-4), (loss, ref)
    dist.barrier()
    if rank == 0:
        shutil.rmtree(tmp, ignore_errors=True)


def test_tp2_pp2_sp_parity():
    _run_workers(_w_tp2_pp2_sp, world_size=4)


def _w_tp4_pp2_sp_70b_dry(rank, world):
    """BASELINE config 3 dry run: TP4 x PP2 + SP at the Llama-3-70B layer
    count (80 layers, tiny hidden) on gloo world 8 — exercises the exact
    process-group layout, stage segmentation and SP-sharded p2p of the
    70B recipe."""
    from paddlenlp_amd.parallel.topology import init_parallel_env
    from paddlenlp_amd.parallel.pipeline import PipelineEngine
    from paddlenlp_amd.transformers import LlamaConfig
    from paddlenlp_amd.transformers.llama.modeling_pp import LlamaForCausalLMPipe

    topo = init_parallel_env(pp_degree=2, mp_degree=4, backend="gloo")
    cfg = LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=80, num_attention_heads=8, num_key_value_heads=4,
        max_position_embeddings=64, dtype="float32",
        tensor_parallel_degree=4, tensor_parallel_rank=topo.get_rank_in("mp"),
        tensor_parallel_output=False, sequence_parallel=True,
    )
    pipe = LlamaForCausalLMPipe(cfg)
    assert len(pipe.local_layers) > 0
    engine = PipelineEngine(pipe, hidden_shape_fn=None, dtype=torch.float32,
                            device=torch.device("cpu"))
    g = torch.Generator().manual_seed(1)
    mbs = [{"input_ids": torch.randint(0, 128, (1, 16), generator=g),
            "labels": torch.randint(0, 128, (1, 16), generator=g)}
           for _ in range(2)]
    loss = engine.forward_backward(mbs, input_fn=lambda mb: mb["input_ids"])
    if pipe.is_last_stage:
        assert torch.isfinite(loss), loss
    # every local parameter received a gradient
    n_grads = sum(1 for p in pipe.parameters() if p.grad is not None)
    assert n_grads > 0
    dist.barrier()


def test_tp4_pp2_sp_70b_dry_run():
    _run_workers(_w_tp4_pp2_sp_70b_dry, world_size=8)
