"""Chinese-CLIP / BiT / ERNIE-Code / BertJapaneseTokenizer.

Reference behavior: paddlenlp/transformers/{chineseclip,bit,ernie_code,
bert_japanese}/.
"""
import torch

from paddlenlp_amd.transformers import (
    BertJapaneseTokenizer,
    BitConfig,
    BitForImageClassification,
    BitModel,
    ChineseCLIPConfig,
    ChineseCLIPModel,
    ErnieCodeConfig,
    ErnieCodeForConditionalGeneration,
)


def test_chineseclip_bert_text_tower_and_contrastive_logits():
    torch.manual_seed(0)
    cfg = ChineseCLIPConfig(
        text_config=dict(vocab_size=96, hidden_size=32, num_hidden_layers=2,
                         num_attention_heads=4, intermediate_size=64),
        vision_config=dict(hidden_size=32, num_hidden_layers=2,
                           num_attention_heads=4, intermediate_size=64,
                           image_size=32, patch_size=8),
        projection_dim=16)
    m = ChineseCLIPModel(cfg).eval()
    ids = torch.randint(0, 96, (3, 10))
    px = torch.randn(2, 3, 32, 32)
    with torch.no_grad():
        lt, li = m(ids, px)
    assert lt.shape == (3, 2) and li.shape == (2, 3)
    # text tower is BERT-style: bidirectional — swapping a later token
    # changes the [CLS] feature (a causal CLIP tower would too, but the
    # reverse direction wouldn't; check future -> CLS flow)
    ids2 = ids.clone()
    ids2[:, -1] = (ids2[:, -1] + 1) % 96
    with torch.no_grad():
        f1 = m.get_text_features(ids)
        f2 = m.get_text_features(ids2)
    assert not torch.allclose(f1, f2, atol=1e-5)


def test_bit_weight_standardized_resnet():
    from paddlenlp_amd.transformers.bit.modeling import WSConv2d

    torch.manual_seed(0)
    cfg = BitConfig(embedding_size=16, hidden_sizes=(16, 32),
                    depths=(1, 1), num_groups=8, num_labels=5)
    m = BitForImageClassification(cfg)
    px = torch.randn(2, 3, 32, 32)
    loss, logits = m(px, labels=torch.tensor([0, 3]))
    assert logits.shape == (2, 5)
    loss.backward()
    # weight standardization: effective conv weights are zero-mean
    conv = m.bit.stages[0][0].conv2
    assert isinstance(conv, WSConv2d)
    w = conv.weight
    mean = w.mean(dim=(1, 2, 3), keepdim=True)
    var = w.var(dim=(1, 2, 3), keepdim=True, unbiased=False)
    ws = (w - mean) * torch.rsqrt(var + 1e-10)
    assert ws.mean(dim=(1, 2, 3)).abs().max() < 1e-5
    # no BatchNorm anywhere (BiT uses GroupNorm only)
    assert not any(isinstance(mod, torch.nn.BatchNorm2d)
                   for mod in m.modules())


def test_ernie_code_is_t5_shaped():
    torch.manual_seed(0)
    cfg = ErnieCodeConfig(vocab_size=96, d_model=32, d_kv=8, d_ff=64,
                          num_layers=2, num_heads=4)
    m = ErnieCodeForConditionalGeneration(cfg)
    src = torch.randint(0, 96, (2, 8))
    labels = torch.randint(0, 96, (2, 6))
    out = m(input_ids=src, labels=labels)
    loss = out[0]
    loss.backward()
    assert float(loss) > 0


def test_bert_japanese_tokenizer(tmp_path):
    vocab = ["[PAD]", "[UNK]", "[CLS]", "[SEP]", "[MASK]",
             "日", "本", "語", "hello", "world", "hel", "##lo"]
    vf = tmp_path / "vocab.txt"
    vf.write_text("\n".join(vocab), encoding="utf-8")
    # character subword mode: unsegmented Japanese -> per-char ids
    tok = BertJapaneseTokenizer(vocab_file=str(vf),
                                subword_tokenizer_type="character")
    ids = tok._tokenizer.encode("日本語").ids
    assert ids == [5, 6, 7]
    # wordpiece mode on spaced text
    tok2 = BertJapaneseTokenizer(vocab_file=str(vf),
                                 subword_tokenizer_type="wordpiece")
    ids2 = tok2._tokenizer.encode("hello world").ids
    assert ids2 == [8, 9]
    # mecab request degrades gracefully
    import warnings

    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        BertJapaneseTokenizer(vocab_file=str(vf),
                              word_tokenizer_type="mecab")
        assert any("MeCab" in str(x.message) for x in w)


# -------------------------------------------------------- speech / audio
def test_speecht5_asr_speech_to_text():
    from paddlenlp_amd.transformers import SpeechT5Config, SpeechT5ForSpeechToText

    torch.manual_seed(0)
    cfg = SpeechT5Config(
        vocab_size=40, hidden_size=32, encoder_layers=2, decoder_layers=2,
        num_attention_heads=4, intermediate_size=64,
        conv_dim=(16, 16), conv_stride=(5, 2), conv_kernel=(10, 3),
        positional_conv_kernel=8, positional_conv_groups=4)
    m = SpeechT5ForSpeechToText(cfg)
    wav = torch.randn(2, 400)                      # raw audio
    labels = torch.randint(3, 40, (2, 6))
    loss, logits = m(wav, labels=labels)
    assert logits.shape == (2, 6, 40)
    loss.backward()
    # the conv feature encoder downsamples the waveform
    frames = m.speecht5.speech_encoder_prenet.feature_encoder(wav).shape[1]
    assert 1 < frames < 400


def test_speecht5_tts_text_to_mel():
    from paddlenlp_amd.transformers import SpeechT5Config, SpeechT5ForTextToSpeech

    torch.manual_seed(0)
    cfg = SpeechT5Config(
        vocab_size=40, hidden_size=32, encoder_layers=2, decoder_layers=2,
        num_attention_heads=4, intermediate_size=64,
        conv_dim=(16,), conv_stride=(5,), conv_kernel=(10,),
        num_mel_bins=20, reduction_factor=2,
        speech_decoder_prenet_units=16, speech_decoder_postnet_units=16,
        positional_conv_kernel=8, positional_conv_groups=4)
    m = SpeechT5ForTextToSpeech(cfg)
    ids = torch.randint(3, 40, (2, 8))
    mel = torch.randn(2, 12, 20)                   # target mel frames
    loss, after = m(ids, labels=mel)
    assert after.shape[0] == 2 and after.shape[2] == 20
    loss.backward()
    # the speech-decoder prenet keeps dropout ON in eval (TTS diversity)
    m.eval()
    with torch.no_grad():
        a1 = m(ids, decoder_mel=mel[:, ::2])
        a2 = m(ids, decoder_mel=mel[:, ::2])
    assert not torch.allclose(a1[1], a2[1])


def test_clap_contrastive_audio_text():
    from paddlenlp_amd.transformers import ClapConfig, ClapModel

    torch.manual_seed(0)
    cfg = ClapConfig(
        audio_config=dict(num_mel_bins=32, max_frames=64, patch_size=8,
                          hidden_size=32, num_hidden_layers=2,
                          num_attention_heads=4, intermediate_size=64),
        text_config=dict(vocab_size=96, hidden_size=32, num_hidden_layers=2,
                         num_attention_heads=4, intermediate_size=64),
        projection_dim=16)
    m = ClapModel(cfg).eval()
    ids = torch.randint(0, 96, (3, 10))
    mel = torch.randn(2, 32, 64)
    with torch.no_grad():
        la, lt = m(ids, mel)
    assert la.shape == (2, 3) and lt.shape == (3, 2)
    # symmetric InfoNCE is trainable end to end
    m.train()
    la, lt = m(ids[:2], mel)
    target = torch.arange(2)
    loss = 0.5 * (torch.nn.functional.cross_entropy(la, target) +
                  torch.nn.functional.cross_entropy(lt, target))
    loss.backward()
    assert m.audio_model.patch_embed.weight.grad is not None


# --------------------------------------------------------- blip_2 / dpt
def test_blip2_qformer_bridge():
    from paddlenlp_amd.transformers import Blip2Config, Blip2Model

    torch.manual_seed(0)
    cfg = Blip2Config(
        vision_config=dict(hidden_size=32, num_hidden_layers=2,
                           num_attention_heads=4, intermediate_size=64,
                           image_size=32, patch_size=8),
        qformer_config=dict(hidden_size=24, num_hidden_layers=4,
                            num_attention_heads=4, intermediate_size=48,
                            cross_attention_frequency=2),
        num_query_tokens=8, lm_hidden_size=40)
    m = Blip2Model(cfg)
    px = torch.randn(2, 3, 32, 32)
    prompt = m(px)
    assert prompt.shape == (2, 8, 40)       # [B, queries, lm_hidden]
    # cross-attention only every 2nd layer
    crosses = [l.cross_attn is not None for l in m.qformer.layers]
    assert crosses == [True, False, True, False]
    # bridge is trainable
    prompt.sum().backward()
    assert m.query_tokens.grad is not None

    emb = torch.randn(2, 5, 40)
    joint = m.generate_inputs_for_lm(px, emb)
    assert joint.shape == (2, 13, 40)


def test_dpt_depth_estimation():
    from paddlenlp_amd.transformers import DPTConfig, DPTForDepthEstimation

    torch.manual_seed(0)
    cfg = DPTConfig(hidden_size=32, num_hidden_layers=4,
                    num_attention_heads=4, intermediate_size=64,
                    image_size=32, patch_size=8,
                    backbone_out_indices=(0, 1, 2, 3),
                    neck_hidden_sizes=(8, 16, 24, 32),
                    fusion_hidden_size=16)
    m = DPTForDepthEstimation(cfg)
    px = torch.randn(2, 3, 32, 32)
    depth = m(px)
    assert depth.dim() == 3 and depth.shape[0] == 2
    loss, _ = m(px, labels=torch.rand(2, 32, 32))
    loss.backward()


# --------------------------------------- multimodal bridge compositions
def _tiny_bridge_kwargs():
    return dict(
        vision_config=dict(hidden_size=32, num_hidden_layers=2,
                           num_attention_heads=4, intermediate_size=64,
                           image_size=32, patch_size=8),
        qformer_config=dict(hidden_size=24, num_hidden_layers=2,
                            num_attention_heads=4, intermediate_size=48),
        num_query_tokens=4)


def test_minigpt4_llama_bridge():
    from paddlenlp_amd.transformers import (
        MiniGPT4Config,
        MiniGPT4ForConditionalGeneration,
    )

    torch.manual_seed(0)
    cfg = MiniGPT4Config(
        text_config=dict(vocab_size=96, hidden_size=32, intermediate_size=64,
                         num_hidden_layers=2, num_attention_heads=4,
                         num_key_value_heads=2, max_position_embeddings=64),
        **_tiny_bridge_kwargs())
    m = MiniGPT4ForConditionalGeneration(cfg)
    px = torch.randn(2, 3, 32, 32)
    ids = torch.randint(0, 96, (2, 6))
    loss, logits = m(px, ids, labels=ids)
    assert logits.shape[1] == 4 + 6          # visual prompt + text
    loss.backward()
    assert m.bridge.query_tokens.grad is not None


def test_visualglm_chatglm_bridge():
    from paddlenlp_amd.transformers import (
        VisualGLMConfig,
        VisualGLMForConditionalGeneration,
    )

    torch.manual_seed(0)
    cfg = VisualGLMConfig(
        text_config=dict(vocab_size=96, hidden_size=32, num_hidden_layers=2,
                         num_attention_heads=4, inner_hidden_size=64,
                         max_sequence_length=64),
        **_tiny_bridge_kwargs())
    m = VisualGLMForConditionalGeneration(cfg)
    px = torch.randn(2, 3, 32, 32)
    ids = torch.randint(0, 96, (2, 6))
    loss, logits = m(px, ids, labels=ids)
    loss.backward()


def test_ernie_vil_contrastive():
    from paddlenlp_amd.transformers import ErnieViLConfig, ErnieViLModel

    torch.manual_seed(0)
    cfg = ErnieViLConfig(
        text_config=dict(vocab_size=96, hidden_size=32, num_hidden_layers=2,
                         num_attention_heads=4, intermediate_size=64),
        vision_config=dict(hidden_size=32, num_hidden_layers=2,
                           num_attention_heads=4, intermediate_size=64,
                           image_size=32, patch_size=8))
    m = ErnieViLModel(cfg).eval()
    with torch.no_grad():
        lt, li = m(torch.randint(0, 96, (3, 8)), torch.randn(2, 3, 32, 32))
    assert lt.shape == (3, 2) and li.shape == (2, 3)


def test_dallebart_text_to_image_tokens():
    from paddlenlp_amd.transformers import (
        DalleBartConfig,
        DalleBartForConditionalGeneration,
    )

    torch.manual_seed(0)
    cfg = DalleBartConfig(
        text_vocab_size=96, image_vocab_size=64, hidden_size=32,
        num_encoder_layers=2, num_decoder_layers=2, num_attention_heads=4,
        intermediate_size=48, max_text_length=16, image_length=9,
        bos_token_id=64)
    m = DalleBartForConditionalGeneration(cfg)
    text = torch.randint(2, 96, (2, 8))
    img = torch.randint(0, 64, (2, 9))       # VQ token grid
    loss, logits = m(text, labels=img)
    assert logits.shape == (2, 9, 65)        # image vocab + BOS
    loss.backward()
    # GLU feed-forward is the dallebart delta
    from paddlenlp_amd.transformers.dallebart.modeling import GLUFeedForward

    assert isinstance(m.dallebart.encoder[0].glu, GLUFeedForward)
    grid = m.generate_image_tokens(text, top_k=8)
    assert grid.shape == (2, 9) and int(grid.max()) < 64


# ------------------------------------ last reference-inventory families
def test_ernie_gen_infill_bias():
    from paddlenlp_amd.transformers import ErnieGenConfig, ErnieGenForGeneration
    from paddlenlp_amd.transformers.ernie_gen.modeling import build_infill_bias

    bias = build_infill_bias(3, 3, torch.device("cpu"), torch.float32)[0, 0]
    neg = torch.finfo(torch.float32).min
    assert bias[0, 2] == 0 and bias[0, 4] == neg   # src sees src, not tgt
    assert bias[4, 1] == 0 and bias[4, 3] == 0      # tgt sees src + past tgt
    assert bias[3, 4] == neg                        # tgt is causal

    torch.manual_seed(0)
    cfg = ErnieGenConfig(vocab_size=96, hidden_size=32, num_hidden_layers=2,
                         num_attention_heads=4, intermediate_size=64,
                         hidden_dropout_prob=0.0)
    m = ErnieGenForGeneration(cfg)
    ids = torch.randint(0, 96, (2, 10))
    labels = ids[:, 6:]
    loss, logits = m(ids, src_len=6, labels=labels)
    assert logits.shape == (2, 4, 96)
    loss.backward()


def test_clipseg_film_segmentation():
    from paddlenlp_amd.transformers import CLIPSegConfig, CLIPSegForImageSegmentation

    torch.manual_seed(0)
    cfg = CLIPSegConfig(
        text_config=dict(vocab_size=96, hidden_size=32, num_hidden_layers=2,
                         num_attention_heads=4, intermediate_size=64),
        vision_config=dict(hidden_size=32, num_hidden_layers=3,
                           num_attention_heads=4, intermediate_size=64,
                           image_size=32, patch_size=8),
        extract_layers=(0, 1, 2), reduce_dim=16, projection_dim=16,
        decoder_intermediate_size=32)
    m = CLIPSegForImageSegmentation(cfg).eval()
    ids = torch.randint(0, 96, (2, 8))
    px = torch.randn(2, 3, 32, 32)
    with torch.no_grad():
        mask_a = m(ids, px)
        # different prompt -> different mask (FiLM conditioning works)
        mask_b = m((ids + 3) % 96, px)
    assert mask_a.dim() == 3
    assert not torch.allclose(mask_a, mask_b, atol=1e-5)
    loss, _ = m(ids, px, labels=torch.randint(0, 2, mask_a.shape).float())
    loss.backward()


def test_semantic_search_dual_and_cross():
    from paddlenlp_amd.transformers import ErnieCrossEncoder, ErnieDualEncoder
    from paddlenlp_amd.transformers.ernie import ErnieConfig

    torch.manual_seed(0)
    cfg = ErnieConfig(vocab_size=96, hidden_size=32, num_hidden_layers=2,
                      num_attention_heads=4, intermediate_size=64)
    dual = ErnieDualEncoder(cfg, output_emb_size=16)
    q = torch.randint(0, 96, (4, 8))
    t = torch.randint(0, 96, (4, 8))
    loss, logits = dual(q, t)
    assert logits.shape == (4, 4)
    loss.backward()
    sim = dual.cosine_sim(q, t)
    assert sim.shape == (4,) and sim.abs().max() <= 1.0 + 1e-5
    # shared towers by default
    assert dual.query_ernie is dual.title_ernie

    cross = ErnieCrossEncoder(cfg)
    loss, logits = cross(q, labels=torch.tensor([0, 1, 0, 1]))
    loss.backward()


def test_layoutlmv2_three_modalities():
    from paddlenlp_amd.transformers import (
        LayoutLMv2Config,
        LayoutLMv2ForTokenClassification,
        LayoutXLMModel,
        LayoutXLMConfig,
    )

    torch.manual_seed(0)
    cfg = LayoutLMv2Config(
        vocab_size=96, hidden_size=32, num_hidden_layers=2,
        num_attention_heads=4, intermediate_size=64,
        image_feature_pool_shape=(4, 4, 32), rel_pos_bins=8,
        max_rel_pos=32, rel_2d_pos_bins=8, max_rel_2d_pos=64,
        num_labels=5, hidden_dropout_prob=0.0)
    m = LayoutLMv2ForTokenClassification(cfg)
    ids = torch.randint(0, 96, (2, 10))
    bbox = torch.randint(0, 1000, (2, 10, 4))
    bbox[:, :, 2:] = bbox[:, :, :2] + 10
    img = torch.randn(2, 3, 32, 32)
    loss, logits = m(ids, bbox, img, labels=torch.randint(0, 5, (2, 10)))
    assert logits.shape == (2, 10, 5)
    loss.backward()

    # bbox reaches the output (spatial embeddings + 2d bias)
    base = m.layoutlmv2
    with torch.no_grad():
        a, _ = base(ids, bbox, img)
        b2 = bbox.clone()
        b2[:, 5] = (b2[:, 5] + 200) % 1000
        b, _ = base(ids, b2, img)
    assert not torch.allclose(a, b, atol=1e-5)

    xlm = LayoutXLMModel(LayoutXLMConfig(
        vocab_size=128, hidden_size=32, num_hidden_layers=1,
        num_attention_heads=4, intermediate_size=64,
        image_feature_pool_shape=(4, 4, 32), rel_pos_bins=8,
        max_rel_pos=32, rel_2d_pos_bins=8, max_rel_2d_pos=64))
    text, vis = xlm(torch.randint(0, 128, (1, 6)))
    assert text.shape == (1, 6, 32) and vis.shape == (1, 16, 32)
