"""Auxiliary subsystems: data-skip replay, memory tracker,
profiler hook, determinism knob.
"""
import json
import os

import pytest
import torch

from paddlenlp_amd.trainer.trainer_utils import (
    TrainerMemoryTracker,
    enable_determinism,
    should_skip_data,
)


def test_should_skip_data():
    assert not should_skip_data(5, None)
    assert not should_skip_data(5, [])
    assert should_skip_data(5, [[3, 7]])
    assert should_skip_data(3, [[3, 7]]) and should_skip_data(7, [[3, 7]])
    assert not should_skip_data(8, [[3, 7]])
    assert should_skip_data(20, [[3, 7], [20, 20]])
    with pytest.raises(ValueError):
        should_skip_data(1, [[9, 3]])


def test_trainer_skips_intervals(tmp_path):
    from paddlenlp_amd.trainer import Trainer
    from paddlenlp_amd.trainer.training_args import TrainingArguments
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=64, hidden_size=32, intermediate_size=64,
                      num_hidden_layers=1, num_attention_heads=2,
                      num_key_value_heads=2, max_position_embeddings=32,
                      dtype="float32")
    model = LlamaForCausalLM.from_config(cfg)
    data = [{"input_ids": torch.randint(0, 64, (8,)),
             "labels": torch.randint(0, 64, (8,))} for _ in range(12)]
    stepped = []
    orig = Trainer.training_step

    def spy(self, m, inputs):
        stepped.append(self.state.global_step + 1)
        return orig(self, m, inputs)

    Trainer.training_step = spy
    try:
        args = TrainingArguments(
            output_dir=str(tmp_path), do_train=True, max_steps=8,
            per_device_train_batch_size=1, logging_steps=1000,
            save_steps=1 << 30, skip_data_intervals=[[2, 3]], report_to=[])
        trainer = Trainer(model=model, args=args, train_dataset=data)
        trainer.train()
    finally:
        Trainer.training_step = orig
    # steps 2 and 3 consumed data but ran no forward/backward
    assert 2 not in stepped and 3 not in stepped
    assert 1 in stepped and 4 in stepped
    assert trainer.state.global_step == 8


def test_memory_tracker():
    m = {}
    t = TrainerMemoryTracker(skip=False)
    t.start("train")
    _ = [bytearray(1 << 20) for _ in range(8)]
    t.stop_and_update_metrics(m)
    assert "train_mem_cpu_rss" in m and m["train_mem_cpu_rss"] > 0
    # skip mode writes nothing
    m2 = {}
    t2 = TrainerMemoryTracker(skip=True)
    t2.start("train")
    t2.stop_and_update_metrics(m2)
    assert m2 == {}


def test_profiler_hook(tmp_path, monkeypatch):
    import paddlenlp_amd.utils.profiler as prof

    trace = tmp_path / "trace.json"
    monkeypatch.setattr(prof, "_profiler", None)
    monkeypatch.setattr(prof, "_batch_range", None)
    monkeypatch.setattr(prof, "_step", 0)
    opts = f"batch_range=[2, 3]; profile_path={trace}"
    x = torch.randn(4, 4)
    for _ in range(5):
        prof.add_profiler_step(opts)
        _ = x @ x
    assert trace.exists()
    data = json.loads(trace.read_text())
    assert "traceEvents" in data


def test_enable_determinism_routes_rms_norm(monkeypatch):
    from paddlenlp_amd import ops

    enable_determinism(7)
    assert os.environ["PNLP_DETERMINISTIC"] == "1"
    a = torch.randn(3, 8)
    b = torch.randn(3, 8)
    torch.manual_seed(7)
    r1 = ops.rms_norm(a, torch.ones(8))
    r2 = ops.rms_norm(a, torch.ones(8))
    torch.testing.assert_close(r1, r2)
    monkeypatch.delenv("PNLP_DETERMINISTIC", raising=False)
    torch.use_deterministic_algorithms(False)


def test_llm_utils_rotary_and_read_res():
    import queue

    from paddlenlp_amd.utils.llm_utils import (
        get_rotary_position_embedding,
        read_res,
    )

    pos = torch.arange(8)[None]
    rot = get_rotary_position_embedding(pos, head_dim=16)
    assert rot.shape == (2, 1, 8, 1, 8)
    torch.testing.assert_close(rot[0, 0, 0, 0], torch.ones(8))  # cos(0)
    # linear scaling halves the effective position
    lin = get_rotary_position_embedding(pos, 16, rope_scaling={"type": "linear", "factor": 2.0})
    torch.testing.assert_close(lin[:, :, 2], rot[:, :, 1])

    class Tok:
        eos_token_id = 99

        def decode(self, ids, skip_special_tokens=True):
            return " ".join(str(i) for i in ids)

    rq, oq = queue.Queue(), queue.Queue()
    rq.put([(0, 5), (1, 7)])
    rq.put([(0, 6), (1, 99)])   # seq 1 ends
    rq.put(None)
    read_res(Tok(), rq, oq)
    results = dict(oq.get_nowait() for _ in range(2))
    assert results[1] == "7"
    assert results[0] == "5 6"   # flushed at stream end


def test_import_utils_and_tools():
    from paddlenlp_amd.utils.import_utils import (
        import_module,
        is_package_available,
    )
    from paddlenlp_amd.utils.tools import compare_version, get_env_device

    assert is_package_available("torch")
    assert not is_package_available("definitely_not_a_module_xyz")
    assert import_module("math") is not None
    assert import_module("not_a_module_abc") is None
    assert get_env_device() in ("cpu", "gpu")
    assert compare_version("2.10.0", "2.9.1") == 1
    assert compare_version("1.0", "1.0.0") == 0
    assert compare_version("1.2", "1.10") == -1


def test_dynamic_loss_scaler():
    from paddlenlp_amd.trainer.trainer_utils import DynamicLossScaler

    s = DynamicLossScaler(init_scale=4.0, growth_interval=2)
    p = torch.nn.Parameter(torch.ones(3))
    loss = (p * 2).sum()
    s.scale_loss(loss).backward()
    torch.testing.assert_close(p.grad, torch.full((3,), 8.0))  # 2 * scale 4
    assert s.unscale_and_check([p])
    torch.testing.assert_close(p.grad, torch.full((3,), 2.0))
    s.update(found_inf=False)
    s.update(found_inf=False)
    assert s.scale == 8.0  # grew after 2 good steps

    p.grad[0] = float("inf")
    assert not s.unscale_and_check([p])
    s.update(found_inf=True)
    assert s.scale == 4.0  # backed off

    sd = s.state_dict()
    s2 = DynamicLossScaler()
    s2.load_state_dict(sd)
    assert s2.scale == s.scale


def test_fp16_trainer_skips_overflow_steps(tmp_path):
    """An inf grad must skip the optimizer step and shrink the scale,
    leaving weights untouched."""
    from paddlenlp_amd.trainer import Trainer
    from paddlenlp_amd.trainer.trainer_utils import DynamicLossScaler
    from paddlenlp_amd.trainer.training_args import TrainingArguments
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=64, hidden_size=32, intermediate_size=64,
                      num_hidden_layers=1, num_attention_heads=2,
                      num_key_value_heads=2, max_position_embeddings=32,
                      dtype="float32")
    model = LlamaForCausalLM.from_config(cfg)
    data = [{"input_ids": torch.randint(0, 64, (8,)),
             "labels": torch.randint(0, 64, (8,))} for _ in range(2)]
    args = TrainingArguments(output_dir=str(tmp_path), do_train=True,
                             max_steps=1, per_device_train_batch_size=1,
                             logging_steps=1000, save_steps=1 << 30,
                             report_to=[])
    trainer = Trainer(model=model, args=args, train_dataset=data)
    trainer.train()  # builds optimizer/scheduler (clean run)

    trainer._loss_scaler = DynamicLossScaler(init_scale=8.0)
    model.zero_grad(set_to_none=True)
    b = {k: v[None] for k, v in data[0].items()}
    # scaled backward, then poison one grad with inf
    trainer.training_step(model, b)
    next(p for p in model.parameters() if p.grad is not None).grad[...] = \
        float("inf")
    w_before = {n: p.detach().clone() for n, p in model.named_parameters()}
    trainer.optimizer_step(model)
    assert trainer._loss_scaler.scale == 4.0  # backed off
    for n, p in model.named_parameters():
        torch.testing.assert_close(p, w_before[n])  # step skipped

    # clean grads step normally and grow the good-step counter
    # (the 1-step linear schedule has decayed lr to 0: restore it)
    for group in trainer.optimizer.param_groups:
        group["lr"] = 1e-3
    trainer.lr_scheduler.step = lambda *a, **k: None  # freeze for the check
    model.zero_grad(set_to_none=True)
    trainer.training_step(model, b)
    trainer.optimizer_step(model)
    changed = any(not torch.equal(p, w_before[n])
                  for n, p in model.named_parameters())
    assert changed


def test_save_strategy_no_never_saves():
    """save_strategy='no' writes no checkpoints even with save_steps set
    — preset-compat guard."""
    import tempfile

    import torch

    from paddlenlp_amd.trainer import Trainer, TrainingArguments
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM
    from torch.utils.data import Dataset

    class DS(Dataset):
        def __len__(self):
            return 8

        def __getitem__(self, i):
            ids = torch.randint(0, 64, (9,),
                                generator=torch.Generator().manual_seed(i))
            return {"input_ids": ids[:-1], "labels": ids[1:]}

    cfg = LlamaConfig(vocab_size=64, hidden_size=32, intermediate_size=64,
                      num_hidden_layers=1, num_attention_heads=4,
                      num_key_value_heads=2, max_position_embeddings=32)
    with tempfile.TemporaryDirectory() as d:
        args = TrainingArguments(output_dir=d, max_steps=4, save_steps=1,
                                 save_strategy="no",
                                 per_device_train_batch_size=2,
                                 logging_steps=100)
        tr = Trainer(model=LlamaForCausalLM.from_config(cfg), args=args,
                     train_dataset=DS())
        tr.train()
        import os

        assert not any(x.startswith("checkpoint-") for x in os.listdir(d))
