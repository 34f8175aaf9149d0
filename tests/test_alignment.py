"""DPO / RM tests."""
import pytest
import torch
from torch.utils.data import Dataset

from paddlenlp_amd.trl import DPOCriterion, DPOTrainer, RewardModel, RewardTrainer, sequence_logprob
from paddlenlp_amd.trainer import TrainingArguments
from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM


def tiny_llama(seed=0):
    torch.manual_seed(seed)
    cfg = LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, dtype="float32",
    )
    return LlamaForCausalLM.from_config(cfg)


class PairDS(Dataset):
    def __init__(self, n=16, s=12, seed=0):
        g = torch.Generator().manual_seed(seed)
        self.chosen = torch.randint(0, 128, (n, s), generator=g)
        self.rejected = torch.randint(0, 128, (n, s), generator=g)

    def __len__(self):
        return len(self.chosen)

    def __getitem__(self, i):
        c, r = self.chosen[i], self.rejected[i]
        cl = c.clone()
        cl[:4] = -100  # mask the "prompt"
        rl = r.clone()
        rl[:4] = -100
        return {
            "chosen_input_ids": c, "chosen_labels": cl,
            "rejected_input_ids": r, "rejected_labels": rl,
        }


@pytest.mark.parametrize("loss_type", ["sigmoid", "hinge", "ipo", "simpo", "dpop", "sppo_hard", "orpo", "kto_pair"])
def test_dpo_criterion_losses(loss_type):
    crit = DPOCriterion(beta=0.1, loss_type=loss_type)
    pc = torch.tensor([-10.0, -12.0])
    pr = torch.tensor([-15.0, -14.0])
    rc = torch.tensor([-11.0, -12.5])
    rr = torch.tensor([-14.0, -13.0])
    loss, cr, rj = crit(pc, pr, rc, rr)
    assert torch.isfinite(loss), loss_type
    assert loss.dim() == 0


def test_dpo_sigmoid_direction():
    """Loss must decrease as the chosen margin grows."""
    crit = DPOCriterion(beta=0.1, loss_type="sigmoid")
    base = crit(torch.tensor([-10.0]), torch.tensor([-10.0]),
                torch.tensor([-10.0]), torch.tensor([-10.0]))[0]
    better = crit(torch.tensor([-8.0]), torch.tensor([-12.0]),
                  torch.tensor([-10.0]), torch.tensor([-10.0]))[0]
    assert better < base


def test_sequence_logprob_masking():
    logits = torch.randn(2, 5, 16)
    labels = torch.full((2, 5), -100, dtype=torch.long)
    labels[:, 2:] = 3
    lp_all = sequence_logprob(logits, labels)
    ref = logits.log_softmax(-1)[:, 2:, 3].sum(-1)
    assert torch.allclose(lp_all, ref, atol=1e-5)


def test_dpo_trainer_step(tmp_path):
    model = tiny_llama()
    args = TrainingArguments(
        output_dir=str(tmp_path), max_steps=3, per_device_train_batch_size=4,
        logging_steps=1, save_steps=1000, learning_rate=1e-4,
    )
    trainer = DPOTrainer(model=model, args=args, train_dataset=PairDS(), beta=0.1)
    out = trainer.train()
    assert out.global_step == 3
    losses = [h["loss"] for h in trainer.state.log_history if "loss" in h]
    assert all(torch.isfinite(torch.tensor(l)) for l in losses)
    # reference model must have stayed frozen
    for p in trainer.reference_model.parameters():
        assert not p.requires_grad


def test_reward_model_step(tmp_path):
    backbone = tiny_llama(seed=2)
    rm = RewardModel(backbone, hidden_size=64)
    args = TrainingArguments(
        output_dir=str(tmp_path), max_steps=3, per_device_train_batch_size=4,
        logging_steps=1, save_steps=1000, learning_rate=1e-4, unified_checkpoint=False,
    )

    class RMDS(PairDS):
        def __getitem__(self, i):
            return {
                "chosen_input_ids": self.chosen[i],
                "rejected_input_ids": self.rejected[i],
            }

    trainer = RewardTrainer(model=rm, args=args, train_dataset=RMDS())
    out = trainer.train()
    assert out.global_step == 3


def test_ppo_step():
    """Two PPO steps on a tiny model: finite losses, actor weights move."""
    import copy

    from paddlenlp_amd.trl import PPOConfig, PPOTrainer, ValueHeadModel

    torch.manual_seed(0)
    actor = tiny_llama(seed=1)
    reference = copy.deepcopy(actor)
    critic = ValueHeadModel(copy.deepcopy(actor), hidden_size=64)

    def reward_fn(prompt, response):
        # toy reward: prefer longer non-pad responses with token diversity
        return float(len(set(response.tolist()))) / max(1, len(response))

    trainer = PPOTrainer(actor, critic, reference, reward_fn,
                         config=PPOConfig(max_new_tokens=8, minibatch_size=2,
                                          ppo_epochs=1))
    before = actor.lm_head.weight.detach().clone()
    prompts = torch.randint(3, 128, (4, 6))
    stats1 = trainer.step(prompts)
    stats2 = trainer.step(prompts)
    for st in (stats1, stats2):
        for k, v in st.items():
            assert abs(v) < 1e6 and v == v, (k, v)  # finite
    assert not torch.equal(before, actor.lm_head.weight)
    # reference stayed frozen
    for p in reference.parameters():
        assert not p.requires_grad


def test_dpo_packed_equals_rowwise():
    """Packed (FlashMask one-row) DPO forward must produce the same loss
    as the batch-concat row form: cross-segment attention is masked and
    RoPE is relative, so the math is identical."""
    import copy

    import torch

    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM
    from paddlenlp_amd.trl.dpo_trainer import DPOTrainer
    from paddlenlp_amd.trainer import TrainingArguments
    import tempfile

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=96, hidden_size=32, intermediate_size=64,
                      num_hidden_layers=2, num_attention_heads=4,
                      num_key_value_heads=2, max_position_embeddings=64,
                      use_flash_attention=True)
    model = LlamaForCausalLM.from_config(cfg)

    ch = torch.randint(2, 96, (2, 10))
    rj = torch.randint(2, 96, (2, 12))
    inputs = {
        "chosen_input_ids": ch,
        "chosen_labels": torch.cat([torch.full((2, 4), -100), ch[:, 4:]],
                                   dim=1),
        "rejected_input_ids": rj,
        "rejected_labels": torch.cat([torch.full((2, 4), -100), rj[:, 4:]],
                                     dim=1),
    }
    with tempfile.TemporaryDirectory() as d:
        args = TrainingArguments(output_dir=d, max_steps=1,
                                 per_device_train_batch_size=2)
        tr = DPOTrainer(model=model, args=args, beta=0.1,
                        loss_type="sigmoid",
                        reference_model=copy.deepcopy(model))
        tr.dpo_packing = True
        packed = tr.compute_loss(model, {k: v.clone()
                                         for k, v in inputs.items()})
        tr.dpo_packing = False
        rowwise = tr.compute_loss(model, {k: v.clone()
                                          for k, v in inputs.items()})
    assert torch.allclose(packed, rowwise, atol=1e-5), \
        (float(packed), float(rowwise))


def test_dpo_packing_adaptive_default():
    """Default packing decision follows the measured crossover: pack only
    when the padded row form wastes >10%."""
    import tempfile

    import torch

    from paddlenlp_amd.trainer import TrainingArguments
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM
    from paddlenlp_amd.trl.dpo_trainer import DPOTrainer

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=96, hidden_size=32, intermediate_size=64,
                      num_hidden_layers=1, num_attention_heads=4,
                      num_key_value_heads=2, max_position_embeddings=128,
                      use_flash_attention=True)
    model = LlamaForCausalLM.from_config(cfg)

    def mk(lc, lr):
        ch = torch.randint(2, 96, (1, lc))
        rj = torch.randint(2, 96, (1, lr))
        return {"chosen_input_ids": ch, "chosen_labels": ch.clone(),
                "rejected_input_ids": rj, "rejected_labels": rj.clone()}

    calls = []
    orig_forward = model.forward

    def spy(*a, **kw):
        calls.append("attn_mask_startend_row_indices" in kw
                     and kw["attn_mask_startend_row_indices"] is not None)
        return orig_forward(*a, **kw)

    model.forward = spy
    with tempfile.TemporaryDirectory() as d:
        args = TrainingArguments(output_dir=d, max_steps=1,
                                 per_device_train_batch_size=1)
        tr = DPOTrainer(model=model, args=args, beta=0.1,
                        loss_type="simpo")   # reference-free: one forward
        # the spy wrapper hides the real signature from the capability
        # probe; assert capability directly
        tr._can_pack = True
        # equal lengths: rowwise (no mask kwarg)
        tr.compute_loss(model, mk(16, 16))
        # heavy mismatch: packed (mask kwarg present)
        tr.compute_loss(model, mk(8, 40))
    assert calls[0] is False and calls[1] is True, calls
