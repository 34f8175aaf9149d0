"""Trainer tests: loop mechanics, checkpoint save/resume loss-equality
(the resume oracle from SURVEY §4.4, single-process form)."""
import json
import os
import tempfile

import pytest
import torch
from torch.utils.data import Dataset

from paddlenlp_amd.trainer import (
    PdArgumentParser,
    Trainer,
    TrainingArguments,
    get_scheduler,
)
from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM


class RandDS(Dataset):
    def __init__(self, n=128, s=32, v=128, seed=0):
        g = torch.Generator().manual_seed(seed)
        self.data = torch.randint(0, v, (n, s + 1), generator=g)

    def __len__(self):
        return len(self.data)

    def __getitem__(self, i):
        return {"input_ids": self.data[i, :-1], "labels": self.data[i, 1:]}


def tiny_model(seed=0):
    torch.manual_seed(seed)
    cfg = LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, dtype="float32",
    )
    return LlamaForCausalLM.from_config(cfg)


def make_args(d, **kw):
    defaults = dict(
        output_dir=d, max_steps=10, per_device_train_batch_size=4,
        gradient_accumulation_steps=1, logging_steps=100, save_steps=1000,
        learning_rate=1e-3, seed=42,
    )
    defaults.update(kw)
    return TrainingArguments(**defaults)


def test_train_decreases_loss():
    """On a memorizable dataset loss must drop."""
    with tempfile.TemporaryDirectory() as d:
        ds = RandDS(n=16, s=16)
        model = tiny_model()
        args = make_args(d, max_steps=60, per_device_train_batch_size=4,
                         learning_rate=3e-3, logging_steps=10)
        tr = Trainer(model=model, args=args, train_dataset=ds)
        out = tr.train()
        first = tr.state.log_history[0]["loss"]
        last = tr.state.log_history[-1]["loss"]
        assert last < first - 0.5, (first, last)


def test_resume_loss_equality():
    """Train 10 steps straight vs 5 + save + resume + 5: identical loss."""
    with tempfile.TemporaryDirectory() as d1, tempfile.TemporaryDirectory() as d2:
        ds = RandDS()
        # straight run
        m1 = tiny_model(seed=7)
        args1 = make_args(d1, max_steps=10, logging_steps=1, save_steps=1000)
        tr1 = Trainer(model=m1, args=args1, train_dataset=ds)
        tr1.train()
        losses1 = [h["loss"] for h in tr1.state.log_history if "loss" in h]

        # interrupted run: same max_steps (same LR schedule), checkpoint at 5;
        # simulate the interrupt by stopping via a callback after step 5
        from paddlenlp_amd.trainer import TrainerCallback

        class StopAt5(TrainerCallback):
            def on_step_end(self, args, state, control, **kw):
                if state.global_step >= 5:
                    control.should_training_stop = True

        m2 = tiny_model(seed=7)
        args2 = make_args(d2, max_steps=10, logging_steps=1, save_steps=5)
        tr2 = Trainer(model=m2, args=args2, train_dataset=ds, callbacks=[StopAt5()])
        tr2.train()
        assert tr2.state.global_step == 5

        m3 = tiny_model(seed=999)  # different init: must be overwritten by ckpt
        args3 = make_args(d2, max_steps=10, logging_steps=1, save_steps=1000)
        tr3 = Trainer(model=m3, args=args3, train_dataset=ds)
        tr3.train(resume_from_checkpoint=os.path.join(d2, "checkpoint-5"))
        losses3 = [h["loss"] for h in tr3.state.log_history if "loss" in h]

        # steps 6..10 of both runs must match to float precision
        # (log_history includes the restored pre-resume entries)
        losses3 = losses3[-5:]
        for a, b in zip(losses1[5:], losses3):
            assert abs(a - b) < 1e-5, (losses1[5:], losses3)


def test_evaluate():
    with tempfile.TemporaryDirectory() as d:
        ds = RandDS()
        model = tiny_model()
        args = make_args(d, max_steps=2)
        tr = Trainer(model=model, args=args, train_dataset=ds, eval_dataset=RandDS(n=16))
        tr.train()
        metrics = tr.evaluate()
        assert "eval_loss" in metrics and metrics["eval_loss"] > 0


def test_lr_schedulers():
    model = torch.nn.Linear(4, 4)
    opt = torch.optim.SGD(model.parameters(), lr=1.0)
    for name in ["linear", "cosine", "constant"]:
        sched = get_scheduler(name, opt, num_warmup_steps=2, num_training_steps=10)
        lrs = []
        for _ in range(10):
            lrs.append(sched.get_last_lr()[0])
            opt.step()
            sched.step()
        assert lrs[0] == 0.0 and lrs[2] > 0.5, (name, lrs)


def test_argparser_json_and_cli(tmp_path):
    cfg = {"output_dir": str(tmp_path), "max_steps": 7, "bf16": True}
    json_file = tmp_path / "cfg.json"
    json_file.write_text(json.dumps(cfg))
    parser = PdArgumentParser(TrainingArguments)
    (args,) = parser.parse_json_file_and_cmd_lines(
        [str(json_file), "--max_steps", "9"])
    assert args.max_steps == 9  # CLI overrides JSON
    assert args.bf16 is True
    assert args.output_dir == str(tmp_path)


def test_save_total_limit():
    with tempfile.TemporaryDirectory() as d:
        ds = RandDS()
        args = make_args(d, max_steps=6, save_steps=2, save_total_limit=2)
        tr = Trainer(model=tiny_model(), args=args, train_dataset=ds)
        tr.train()
        ckpts = sorted(x for x in os.listdir(d) if x.startswith("checkpoint"))
        assert ckpts == ["checkpoint-4", "checkpoint-6"]


def test_reference_config_strings_parse():
    """Reference-style recipe JSON (tp/pp/sharding config strings) parses
    and maps onto the native switches (reference training_args.py:655-690)."""
    import json
    import tempfile

    from paddlenlp_amd.trainer import TrainingArguments
    from paddlenlp_amd.trainer.argparser import PdArgumentParser

    recipe = {
        "output_dir": "out",
        "per_device_train_batch_size": 1,
        "tensor_parallel_config": "enable_mp_async_allreduce enable_mp_skip_c_identity",
        "pipeline_parallel_config": "enable_timer disable_p2p_cache_shape",
        "sharding_parallel_config": "enable_stage1_tensor_fusion enable_stage1_overlap split_param",
        "scale_loss": 1024.0,
        "fp16_opt_level": "O2",
        "hybrid_parallel_topo_order": "sharding_first",
    }
    with tempfile.NamedTemporaryFile("w", suffix=".json", delete=False) as f:
        json.dump(recipe, f)
        path = f.name
    parser = PdArgumentParser([TrainingArguments])
    (args,) = parser.parse_json_file(path)
    assert "enable_mp_async_allreduce" in args.tp_options
    assert args.skip_profile_timer is False          # enable_timer
    assert args.sharding_overlap_comm is True        # enable_stage1_overlap
    assert args.scale_loss == 1024.0


def test_load_best_model_at_end():
    """Best checkpoint (by eval_loss) is tracked, survives rotation, and is
    reloaded at train end (reference trainer.py:2464-2516)."""
    with tempfile.TemporaryDirectory() as d:
        ds = RandDS(n=32, s=16)
        model = tiny_model()
        args = make_args(
            d, max_steps=12, save_steps=4, eval_steps=4,
            evaluation_strategy="steps", load_best_model_at_end=True,
            metric_for_best_model="eval_loss", save_total_limit=1,
            learning_rate=5e-3,
        )
        tr = Trainer(model=model, args=args, train_dataset=ds,
                     eval_dataset=RandDS(n=16, s=16, seed=1))
        tr.train()
        best = tr.state.best_model_checkpoint
        assert best is not None and os.path.isdir(best), best
        assert tr.state.best_metric is not None
        # model weights must equal the best checkpoint's saved weights
        from safetensors.torch import load_file

        path = os.path.join(best, "model.safetensors")
        saved = load_file(path)
        live = {k: v for k, v in model.state_dict().items()}
        for k, v in saved.items():
            assert torch.equal(v, live[k].to(v.dtype)), k


def test_log_save_metrics_and_state():
    with tempfile.TemporaryDirectory() as d:
        tr = Trainer(model=tiny_model(), args=make_args(d),
                     train_dataset=RandDS(n=8))
        m = {"eval_loss": 1.5, "eval_accuracy": 0.75}
        tr.log_metrics("eval", m)
        tr.save_metrics("eval", m)
        tr.save_metrics("train", {"train_loss": 2.0})
        with open(os.path.join(d, "eval_results.json")) as f:
            assert json.load(f) == m
        with open(os.path.join(d, "all_results.json")) as f:
            merged = json.load(f)
        assert merged["eval_loss"] == 1.5 and merged["train_loss"] == 2.0
        tr.save_state()
        assert os.path.isfile(os.path.join(d, "trainer_state.json"))


def test_epoch_strategies_with_load_best():
    """The SFT-preset pattern: evaluation_strategy=save_strategy=epoch +
    load_best_model_at_end works end to end."""
    with tempfile.TemporaryDirectory() as d:
        ds = RandDS(n=24, s=16)
        model = tiny_model()
        args = make_args(
            d, max_steps=-1, num_train_epochs=3, save_steps=0,
            evaluation_strategy="epoch", save_strategy="epoch",
            load_best_model_at_end=True, metric_for_best_model="eval_loss",
            save_total_limit=2, learning_rate=3e-3,
            per_device_train_batch_size=4)
        tr = Trainer(model=model, args=args, train_dataset=ds,
                     eval_dataset=RandDS(n=8, s=16, seed=3))
        tr.train()
        # 3 epochs -> 3 evals recorded; best checkpoint exists and loaded
        evals = [l for l in tr.state.log_history if "eval_loss" in l]
        assert len(evals) == 3, tr.state.log_history
        assert tr.state.best_model_checkpoint is not None
        assert os.path.isdir(tr.state.best_model_checkpoint)


def test_load_best_by_custom_metric():
    """metric_for_best_model='accuracy' (greater_is_better inferred True)
    tracks the best checkpoint via compute_metrics output."""
    with tempfile.TemporaryDirectory() as d:
        ds = RandDS(n=16, s=16)
        vals = iter([0.2, 0.8, 0.5])

        def metrics_fn(pack):
            return {"accuracy": next(vals)}

        args = make_args(d, max_steps=6, save_steps=2, eval_steps=2,
                         evaluation_strategy="steps",
                         load_best_model_at_end=True,
                         metric_for_best_model="accuracy")
        tr = Trainer(model=tiny_model(), args=args, train_dataset=ds,
                     eval_dataset=RandDS(n=8, s=16, seed=2),
                     compute_metrics=metrics_fn)
        tr.train()
        assert tr.state.best_metric == 0.8
        assert tr.state.best_model_checkpoint.endswith("checkpoint-4")
