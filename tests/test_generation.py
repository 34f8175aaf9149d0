"""Generation tests: greedy/sample/beam, KV cache equivalence, processors."""
import pytest
import torch

from paddlenlp_amd.generation import GenerationConfig
from paddlenlp_amd.generation.logits_process import (
    TopKLogitsWarper,
    TopPLogitsWarper,
)
from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM


@pytest.fixture(scope="module")
def model():
    torch.manual_seed(0)
    cfg = LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=256, eos_token_id=2, pad_token_id=0,
    )
    m = LlamaForCausalLM.from_config(cfg)
    m.eval()
    return m


def test_greedy_cached_equals_uncached(model):
    ids = torch.randint(3, 128, (2, 8), generator=torch.Generator().manual_seed(1))
    out_cached, _ = model.generate(ids, max_new_tokens=12, use_cache=True)
    out_full, _ = model.generate(ids, max_new_tokens=12, use_cache=False)
    assert torch.equal(out_cached, out_full)


def test_sample_deterministic_with_seed(model):
    ids = torch.randint(3, 128, (1, 8), generator=torch.Generator().manual_seed(1))
    torch.manual_seed(7)
    out1, _ = model.generate(ids, max_new_tokens=10, do_sample=True, top_k=20)
    torch.manual_seed(7)
    out2, _ = model.generate(ids, max_new_tokens=10, do_sample=True, top_k=20)
    assert torch.equal(out1, out2)


def test_beam_search_runs(model):
    ids = torch.randint(3, 128, (2, 6), generator=torch.Generator().manual_seed(2))
    out, _ = model.generate(ids, max_new_tokens=8, num_beams=3)
    assert out.shape[0] == 2 and out.shape[1] <= 8
    # beam score of returned seq should be >= greedy's
    greedy, _ = model.generate(ids, max_new_tokens=8)

    def seq_logprob(prompt, cont):
        full = torch.cat([prompt, cont], dim=-1)
        logits = model(input_ids=full)
        lp = logits[:, :-1].log_softmax(-1)
        tgt = full[:, 1:]
        tok_lp = lp.gather(-1, tgt[..., None]).squeeze(-1)
        return tok_lp[:, prompt.shape[1] - 1:].sum(-1)

    if out.shape == greedy.shape:
        assert (seq_logprob(ids, out) >= seq_logprob(ids, greedy) - 1e-4).all()


def test_top_k_warper():
    logits = torch.tensor([[1.0, 5.0, 3.0, 2.0]])
    out = TopKLogitsWarper(2)(None, logits.clone())
    assert out[0, 0] == float("-inf") and out[0, 3] == float("-inf")
    assert out[0, 1] == 5.0 and out[0, 2] == 3.0


def test_top_p_warper():
    logits = torch.log(torch.tensor([[0.5, 0.3, 0.15, 0.05]]))
    out = TopPLogitsWarper(0.7)(None, logits.clone())
    probs = out.softmax(-1)
    assert probs[0, 3] == 0.0 and probs[0, 0] > 0


def test_generation_config_roundtrip(tmp_path):
    gen = GenerationConfig(max_new_tokens=77, top_p=0.9, eos_token_id=[1, 2])
    gen.save_pretrained(str(tmp_path))
    loaded = GenerationConfig.from_pretrained(str(tmp_path))
    assert loaded.max_new_tokens == 77 and loaded.eos_token_id == [1, 2]


def test_eos_stopping(model):
    """Zero the EOS row and add a huge bias via the weight so EOS wins
    regardless of the hidden state's sign."""
    with torch.no_grad():
        saved = model.lm_head.weight[2].clone()
        model.lm_head.weight[2] = 0.0
    # monkeypatch forward-time bias through a hook on lm_head
    handle = model.lm_head.register_forward_hook(
        lambda mod, inp, out: out.index_fill(-1, torch.tensor(2), 1000.0))
    ids = torch.randint(3, 128, (2, 4))
    out, _ = model.generate(ids, max_new_tokens=20)
    handle.remove()
    with torch.no_grad():
        model.lm_head.weight[2] = saved
    assert out.shape[1] < 20  # stopped early at EOS
    assert (out[:, 0] == 2).all()


def test_group_beam_search_diversity(model):
    """Group beam search runs, and with a strong diversity penalty the two
    groups pick different first tokens."""
    ids = torch.randint(3, 128, (2, 6))
    gen = GenerationConfig(max_new_tokens=6, num_beams=4, num_beam_groups=2,
                           diversity_penalty=0.0, eos_token_id=2, pad_token_id=0)
    out, _ = model.generate(ids, gen)
    assert out.shape[0] == 2 and out.shape[1] <= 6

    # plain beam search and group beam search with zero penalty agree on the
    # best sequence (same search, grouped)
    plain_gen = GenerationConfig(max_new_tokens=6, num_beams=2, eos_token_id=2,
                                 pad_token_id=0)
    plain, _ = model.generate(ids, plain_gen)
    gen0 = GenerationConfig(max_new_tokens=6, num_beams=4, num_beam_groups=2,
                            diversity_penalty=0.0, eos_token_id=2, pad_token_id=0)
    grouped, _ = model.generate(ids, gen0)
    L = min(plain.shape[1], grouped.shape[1])
    assert torch.equal(plain[:, :L], grouped[:, :L])


def test_hamming_diversity_processor():
    from paddlenlp_amd.generation.logits_process import (
        HammingDiversityLogitsProcessor,
    )

    proc = HammingDiversityLogitsProcessor(
        diversity_penalty=1.5, num_beams=4, num_beam_groups=2)
    scores = torch.zeros(4, 10)  # B=2 x Kg=2
    used = torch.zeros(2, 10)
    used[0, 3] = 2.0
    used[1, 7] = 1.0
    out = proc(scores, used)
    assert out[0, 3] == -3.0 and out[1, 3] == -3.0   # batch 0 rows
    assert out[2, 7] == -1.5 and out[3, 7] == -1.5   # batch 1 rows
    assert out[0, 0] == 0.0


def test_speculative_greedy_matches_target_greedy(model):
    """Greedy speculative decode must be IDENTICAL to target-only greedy."""
    from paddlenlp_amd.generation.speculative import speculative_generate

    torch.manual_seed(1)
    draft_cfg = LlamaConfig(
        vocab_size=128, hidden_size=32, intermediate_size=64,
        num_hidden_layers=1, num_attention_heads=2, num_key_value_heads=2,
        max_position_embeddings=256, eos_token_id=2, pad_token_id=0,
    )
    draft = LlamaForCausalLM(draft_cfg).eval()

    for seed in (0, 3):
        g = torch.Generator().manual_seed(seed)
        ids = torch.randint(3, 128, (1, 8), generator=g)
        gen = GenerationConfig(max_new_tokens=12, do_sample=False,
                               eos_token_id=2, pad_token_id=0)
        ref, _ = model.generate(ids, gen)
        spec, stats = speculative_generate(model, draft, ids, gen, gamma=3)
        L = min(ref.shape[1], spec.shape[1])
        assert torch.equal(ref[0, :L], spec[0, :L]), (ref, spec)
        assert stats["drafted"] > 0

    # a perfect draft (the target itself) accepts everything
    _, stats = speculative_generate(model, model, ids, gen, gamma=3)
    assert stats["acceptance_rate"] == 1.0


def test_speculative_sampling_runs(model):
    from paddlenlp_amd.generation.speculative import speculative_generate

    torch.manual_seed(2)
    ids = torch.randint(3, 128, (1, 6))
    gen = GenerationConfig(max_new_tokens=8, do_sample=True, temperature=1.0,
                           eos_token_id=2, pad_token_id=0)
    out, stats = speculative_generate(model, model, ids, gen, gamma=2)
    assert out.shape[1] <= 8
    assert 0.0 <= stats["acceptance_rate"] <= 1.0


def test_stopping_criteria(model):
    import time

    from paddlenlp_amd.generation import (
        MaxLengthCriteria,
        MaxNewTokensCriteria,
        MaxTimeCriteria,
        StoppingCriteriaList,
    )

    ids = torch.randint(3, 128, (1, 6))
    crit = StoppingCriteriaList([MaxLengthCriteria(10)])
    gen = GenerationConfig(max_new_tokens=20, do_sample=False,
                           eos_token_id=None, pad_token_id=0)
    out, _ = model.greedy_search(ids, gen, stopping_criteria=crit)
    assert out.shape[1] == 4  # stopped at total length 10

    assert MaxNewTokensCriteria(6, 3)(torch.zeros(1, 9), None)
    assert not MaxNewTokensCriteria(6, 3)(torch.zeros(1, 8), None)
    t = MaxTimeCriteria(max_time=1000.0, initial_timestamp=time.time())
    assert not t(torch.zeros(1, 1), None)
    t2 = MaxTimeCriteria(max_time=0.0, initial_timestamp=time.time() - 1)
    assert t2(torch.zeros(1, 1), None)


def test_num_return_sequences(model):
    ids = torch.randint(3, 128, (2, 5))
    # beam: top-2 sequences per batch item, flattened [B*n, L]
    gen = GenerationConfig(max_new_tokens=5, num_beams=4,
                           num_return_sequences=2, eos_token_id=2,
                           pad_token_id=0)
    out, _ = model.generate(ids, gen)
    assert out.shape[0] == 4
    # the first sequence per item is the best (same as n=1 beam search)
    gen1 = GenerationConfig(max_new_tokens=5, num_beams=4, eos_token_id=2,
                            pad_token_id=0)
    best, _ = model.generate(ids, gen1)
    L = min(best.shape[1], out.shape[1])
    assert torch.equal(out[0, :L], best[0, :L])
    assert torch.equal(out[2, :L], best[1, :L])

    # sampling: independent expansions
    torch.manual_seed(0)
    gen_s = GenerationConfig(max_new_tokens=4, do_sample=True,
                             num_return_sequences=3, eos_token_id=2,
                             pad_token_id=0)
    out_s, _ = model.generate(ids, gen_s)
    assert out_s.shape[0] == 6

    # greedy with n>1 is rejected (reference semantics)
    import pytest as _pytest

    with _pytest.raises(ValueError):
        model.generate(ids, GenerationConfig(max_new_tokens=2,
                                             num_return_sequences=2))


def test_no_repeat_ngram_and_forced_tokens():
    from paddlenlp_amd.generation.logits_process import (
        ForcedBOSTokenLogitsProcessor,
        ForcedEOSTokenLogitsProcessor,
        NoRepeatNGramLogitsProcessor,
    )

    ids = torch.tensor([[1, 2, 3, 1, 2]])
    logits = torch.zeros(1, 10)
    out = NoRepeatNGramLogitsProcessor(3)(ids, logits.clone())
    # prefix (1,2) previously continued with 3 -> 3 banned
    assert out[0, 3] == -float("inf") and out[0, 4] == 0

    bos = ForcedBOSTokenLogitsProcessor(prompt_len=5, bos_token_id=7)
    out = bos(ids, torch.zeros(1, 10))
    assert out[0, 7] == 0 and out[0, 0] == -float("inf")

    eos = ForcedEOSTokenLogitsProcessor(max_total_len=6, eos_token_id=9)
    out = eos(ids, torch.zeros(1, 10))
    assert out[0, 9] == 0 and out[0, 1] == -float("inf")


def test_bad_words_and_sequence_bias():
    from paddlenlp_amd.generation.logits_process import (
        NoBadWordsLogitsProcessor,
        SequenceBiasLogitsProcessor,
    )

    ids = torch.tensor([[4, 5], [5, 6]])
    logits = torch.zeros(2, 10)
    # ban token 2 outright; ban "6 then 7" as a sequence
    out = NoBadWordsLogitsProcessor([[2], [6, 7]])(ids, logits.clone())
    assert (out[:, 2] == -float("inf")).all()
    assert out[0, 7] == 0                    # row 0 tail is 5, allowed
    assert out[1, 7] == -float("inf")        # row 1 tail is 6, banned

    out = SequenceBiasLogitsProcessor({(3,): 2.5})(ids, torch.zeros(2, 10))
    assert (out[:, 3] == 2.5).all()


def test_prefix_constrained_generation():
    from paddlenlp_amd.generation.logits_process import (
        PrefixConstrainedLogitsProcessor,
    )

    proc = PrefixConstrainedLogitsProcessor(lambda b, ids: [1, 2], 1)
    out = proc(torch.tensor([[0]]), torch.zeros(1, 5))
    assert out[0, 1] == 0 and out[0, 3] == -float("inf")


def test_generate_with_no_repeat_ngram(model):
    torch.manual_seed(0)
    ids = torch.randint(0, 96, (1, 6))
    out, _ = model.generate(
        ids, GenerationConfig(max_new_tokens=12, do_sample=False,
                              no_repeat_ngram_size=2, pad_token_id=0))
    toks = out[0].tolist()
    seen = set()
    full = ids[0].tolist() + toks
    ok = True
    for i in range(len(full) - 1):
        pair = (full[i], full[i + 1])
        if pair in seen:
            ok = False
        seen.add(pair)
    assert ok, full


def test_forced_bos_in_generate(model):
    """forced_bos_token_id pins the first generated token (the mBART
    translation convention)."""
    torch.manual_seed(0)
    ids = torch.randint(0, 96, (2, 5))
    out, _ = model.generate(
        ids, GenerationConfig(max_new_tokens=4, do_sample=False,
                              forced_bos_token_id=7, pad_token_id=0))
    assert (out[:, 0] == 7).all()


def test_bad_words_never_generated(model):
    torch.manual_seed(0)
    ids = torch.randint(0, 96, (1, 5))
    # ban whatever greedy would pick first
    base, _ = model.generate(ids, GenerationConfig(max_new_tokens=1,
                                                   do_sample=False,
                                                   pad_token_id=0))
    banned = int(base[0, 0])
    out, _ = model.generate(
        ids, GenerationConfig(max_new_tokens=6, do_sample=False,
                              bad_words_ids=[[banned]], pad_token_id=0))
    assert banned not in out[0].tolist()


def test_generation_config_roundtrip(tmp_path):
    """save_pretrained drops non-serializable fields and round-trips the
    rest."""
    from paddlenlp_amd.generation import GenerationConfig

    g = GenerationConfig(max_new_tokens=9, no_repeat_ngram_size=3,
                         bad_words_ids=[[1, 2]], forced_eos_token_id=5,
                         prefix_allowed_tokens_fn=lambda b, ids: [0])
    g.save_pretrained(str(tmp_path))
    loaded = GenerationConfig.from_pretrained(str(tmp_path))
    assert loaded.max_new_tokens == 9
    assert loaded.no_repeat_ngram_size == 3
    assert loaded.bad_words_ids == [[1, 2]]
    assert loaded.forced_eos_token_id == 5
    assert loaded.prefix_allowed_tokens_fn is None
