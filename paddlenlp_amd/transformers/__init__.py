from .configuration_utils import LlmMetaConfig, PretrainedConfig  # noqa: F401
from .model_utils import PretrainedModel, unwrap_model  # noqa: F401
from .tokenizer_utils import PretrainedTokenizer  # noqa: F401
from .auto import (  # noqa: F401
    AutoConfig,
    AutoModel,
    AutoModelForCausalLM,
    AutoModelForConditionalGeneration,
    AutoModelForSeq2SeqLM,
    AutoTokenizer,
)
from .llama import (  # noqa: F401
    LlamaConfig,
    LlamaForCausalLM,
    LlamaModel,
    LlamaPretrainingCriterion,
)
from .gpt import GPTConfig, GPTForCausalLM, GPTModel  # noqa: F401
from .bert import (  # noqa: F401
    BertConfig,
    BertForMaskedLM,
    BertForMultipleChoice,
    BertForPretraining,
    BertForQuestionAnswering,
    BertForSequenceClassification,
    BertForTokenClassification,
    BertModel,
)
from .ernie import (  # noqa: F401
    UIE,
    UTC,
    ErnieConfig,
    ErnieForMaskedLM,
    ErnieForPretraining,
    ErnieForQuestionAnswering,
    ErnieForSequenceClassification,
    ErnieForTokenClassification,
    ErnieModel,
)
from .roberta import (  # noqa: F401
    RobertaConfig,
    RobertaForMaskedLM,
    RobertaForQuestionAnswering,
    RobertaForSequenceClassification,
    RobertaForTokenClassification,
    RobertaModel,
)
from .electra import (  # noqa: F401
    ElectraConfig,
    ElectraDiscriminator,
    ElectraForSequenceClassification,
    ElectraForTokenClassification,
    ElectraForTotalPretraining,
    ElectraGenerator,
    ElectraModel,
)
from .t5 import (  # noqa: F401
    T5Config,
    T5EncoderModel,
    T5ForConditionalGeneration,
    T5Model,
)
from .qwen2_moe import (  # noqa: F401
    Qwen2MoeConfig,
    Qwen2MoeForCausalLM,
    Qwen2MoeModel,
)
from .deepseek_v2 import (  # noqa: F401
    DeepseekV2Config,
    DeepseekV2ForCausalLM,
    DeepseekV2Model,
)
from .bart import (  # noqa: F401
    BartConfig,
    BartForConditionalGeneration,
    BartModel,
)
from .pegasus import (  # noqa: F401
    PegasusConfig,
    PegasusForConditionalGeneration,
    PegasusModel,
)
from .mbart import (  # noqa: F401
    MBartConfig,
    MBartForConditionalGeneration,
    MBartModel,
)
from .qwen import (  # noqa: F401
    QWenConfig,
    QWenForCausalLM,
    QWenLMHeadModel,
    QWenModel,
)
from .codegen import (  # noqa: F401
    CodeGenConfig,
    CodeGenForCausalLM,
    CodeGenModel,
)
from .tinybert import (  # noqa: F401
    TinyBertConfig,
    TinyBertForPretraining,
    TinyBertForQuestionAnswering,
    TinyBertForSequenceClassification,
    TinyBertModel,
)
from .ppminilm import (  # noqa: F401
    PPMiniLMConfig,
    PPMiniLMForQuestionAnswering,
    PPMiniLMForSequenceClassification,
    PPMiniLMModel,
)
from .skep import (  # noqa: F401
    SkepConfig,
    SkepCrfForTokenClassification,
    SkepForSequenceClassification,
    SkepForTokenClassification,
    SkepModel,
)
from .yuan import YuanConfig, YuanForCausalLM, YuanModel  # noqa: F401
from .blenderbot import (  # noqa: F401
    BlenderbotConfig,
    BlenderbotForConditionalGeneration,
    BlenderbotModel,
)
from .blenderbot_small import (  # noqa: F401
    BlenderbotSmallConfig,
    BlenderbotSmallForConditionalGeneration,
    BlenderbotSmallModel,
)
from .nezha import (  # noqa: F401
    NeZhaConfig,
    NeZhaForQuestionAnswering,
    NeZhaForSequenceClassification,
    NeZhaForTokenClassification,
    NeZhaModel,
)
from .mpnet import (  # noqa: F401
    MPNetConfig,
    MPNetForMaskedLM,
    MPNetForSequenceClassification,
    MPNetModel,
)
from .fnet import (  # noqa: F401
    FNetConfig,
    FNetForMaskedLM,
    FNetForSequenceClassification,
    FNetModel,
)
from .ernie_gram import (  # noqa: F401
    ErnieGramConfig,
    ErnieGramForQuestionAnswering,
    ErnieGramForSequenceClassification,
    ErnieGramForTokenClassification,
    ErnieGramModel,
)
from .rembert import (  # noqa: F401
    RemBertConfig,
    RemBertForMaskedLM,
    RemBertForSequenceClassification,
    RemBertModel,
)
from .xlm import (  # noqa: F401
    XLMConfig,
    XLMForSequenceClassification,
    XLMModel,
    XLMWithLMHeadModel,
)
from .unified_transformer import (  # noqa: F401
    UnifiedTransformerConfig,
    UnifiedTransformerLMHeadModel,
    UnifiedTransformerModel,
)
from .unimo import UNIMOConfig, UNIMOLMHeadModel, UNIMOModel  # noqa: F401
from .chatglm import (  # noqa: F401
    ChatGLMConfig,
    ChatGLMForCausalLM,
    ChatGLMModel,
)
from .megatronbert import (  # noqa: F401
    MegatronBertConfig,
    MegatronBertForQuestionAnswering,
    MegatronBertForSequenceClassification,
    MegatronBertModel,
)
from .layoutlm import (  # noqa: F401
    LayoutLMConfig,
    LayoutLMForSequenceClassification,
    LayoutLMForTokenClassification,
    LayoutLMModel,
)
from .distill_utils import (  # noqa: F401
    calc_minilm_loss,
    calc_multi_relation_loss,
    to_distill,
)
from .gemma import GemmaConfig, GemmaForCausalLM, GemmaModel  # noqa: F401
from .opt import OPTConfig, OPTForCausalLM, OPTModel  # noqa: F401
from .bloom import BloomConfig, BloomForCausalLM, BloomModel  # noqa: F401
from .falcon import FalconConfig, FalconForCausalLM, FalconModel  # noqa: F401
from .chatglm_v2 import (  # noqa: F401
    ChatGLMv2Config,
    ChatGLMv2ForCausalLM,
    ChatGLMv2Model,
)
from .mamba import MambaConfig, MambaForCausalLM, MambaModel  # noqa: F401
from .gptj import GPTJConfig, GPTJForCausalLM, GPTJModel  # noqa: F401
from .albert import (  # noqa: F401
    AlbertConfig,
    AlbertForMaskedLM,
    AlbertForSequenceClassification,
    AlbertModel,
)
from .clip import (  # noqa: F401
    CLIPConfig,
    CLIPModel,
    CLIPTextModel,
    CLIPVisionModel,
)
from .ernie_layout import (  # noqa: F401
    ErnieLayoutConfig,
    ErnieLayoutForQuestionAnswering,
    ErnieLayoutForTokenClassification,
    ErnieLayoutModel,
)
from .jamba import JambaConfig, JambaForCausalLM, JambaModel  # noqa: F401
from .qwen2 import Qwen2Config, Qwen2ForCausalLM, Qwen2Model  # noqa: F401
from .mistral import (  # noqa: F401
    MistralConfig,
    MistralForCausalLM,
    MistralModel,
)
from .mixtral import (  # noqa: F401
    MixtralConfig,
    MixtralForCausalLM,
    MixtralModel,
)
from .ernie_m import (  # noqa: F401
    ErnieMConfig,
    ErnieMForSequenceClassification,
    ErnieMForTokenClassification,
    ErnieMModel,
)
from .blip import (  # noqa: F401
    BlipConfig,
    BlipForConditionalGeneration,
    BlipForImageTextRetrieval,
    BlipModel,
)
from .distilbert import (  # noqa: F401
    DistilBertConfig,
    DistilBertForMaskedLM,
    DistilBertForSequenceClassification,
    DistilBertModel,
)
from .roformer import (  # noqa: F401
    RoFormerConfig,
    RoFormerForMaskedLM,
    RoFormerForSequenceClassification,
    RoFormerModel,
)
from .deberta import (  # noqa: F401
    DebertaConfig,
    DebertaForMaskedLM,
    DebertaForSequenceClassification,
    DebertaModel,
)
from .mt5 import (  # noqa: F401
    MT5Config,
    MT5EncoderModel,
    MT5ForConditionalGeneration,
    MT5Model,
)
from .xlnet import (  # noqa: F401
    XLNetConfig,
    XLNetForSequenceClassification,
    XLNetLMHeadModel,
    XLNetModel,
)
from .reformer import (  # noqa: F401
    ReformerConfig,
    ReformerModel,
    ReformerModelWithLMHead,
)
from .bigbird import (  # noqa: F401
    BigBirdConfig,
    BigBirdForMaskedLM,
    BigBirdForSequenceClassification,
    BigBirdModel,
)
from .nystromformer import (  # noqa: F401
    NystromformerConfig,
    NystromformerForSequenceClassification,
    NystromformerModel,
)
from .convbert import (  # noqa: F401
    ConvBertConfig,
    ConvBertForMaskedLM,
    ConvBertForSequenceClassification,
    ConvBertModel,
)
from .ctrl import (  # noqa: F401
    CTRLConfig,
    CTRLForSequenceClassification,
    CTRLLMHeadModel,
    CTRLModel,
)
from .glm import (  # noqa: F401
    GLMConfig,
    GLMForConditionalGeneration,
    GLMModel,
)
from .mobilebert import (  # noqa: F401
    MobileBertConfig,
    MobileBertForSequenceClassification,
    MobileBertModel,
)
from .squeezebert import (  # noqa: F401
    SqueezeBertConfig,
    SqueezeBertForSequenceClassification,
    SqueezeBertModel,
)
from .gau_alpha import (  # noqa: F401
    GAUAlphaConfig,
    GAUAlphaForSequenceClassification,
    GAUAlphaModel,
)
from .deberta_v2 import (  # noqa: F401
    DebertaV2Config,
    DebertaV2ForMaskedLM,
    DebertaV2ForSequenceClassification,
    DebertaV2Model,
)
from .chinesebert import (  # noqa: F401
    ChineseBertConfig,
    ChineseBertForSequenceClassification,
    ChineseBertModel,
)
from .funnel import (  # noqa: F401
    FunnelConfig,
    FunnelForSequenceClassification,
    FunnelForTokenClassification,
    FunnelModel,
)
from .prophetnet import (  # noqa: F401
    ProphetNetConfig,
    ProphetNetForConditionalGeneration,
    ProphetNetModel,
)
from .luke import (  # noqa: F401
    LukeConfig,
    LukeForEntityClassification,
    LukeModel,
)
from .roformerv2 import (  # noqa: F401
    RoFormerv2Config,
    RoFormerv2ForSequenceClassification,
    RoFormerv2Model,
)
from .ernie_ctm import (  # noqa: F401
    ErnieCtmConfig,
    ErnieCtmModel,
    ErnieCtmWordtagModel,
)
from .ernie_doc import (  # noqa: F401
    ErnieDocConfig,
    ErnieDocForSequenceClassification,
    ErnieDocModel,
)
from .transformer import (  # noqa: F401
    TransformerConfig,
    TransformerModel,
)
from .chineseclip import (  # noqa: F401
    ChineseCLIPConfig,
    ChineseCLIPModel,
)
from .bit import (  # noqa: F401
    BitConfig,
    BitForImageClassification,
    BitModel,
)
from .ernie_code import (  # noqa: F401
    ErnieCodeConfig,
    ErnieCodeEncoderModel,
    ErnieCodeForConditionalGeneration,
    ErnieCodeModel,
)
from .bert_japanese import BertJapaneseTokenizer  # noqa: F401
from .speecht5 import (  # noqa: F401
    SpeechT5Config,
    SpeechT5ForSpeechToText,
    SpeechT5ForTextToSpeech,
    SpeechT5Model,
)
from .clap import (  # noqa: F401
    ClapConfig,
    ClapModel,
)
from .blip_2 import (  # noqa: F401
    Blip2Config,
    Blip2Model,
    Blip2QFormerModel,
)
from .dpt import (  # noqa: F401
    DPTConfig,
    DPTForDepthEstimation,
    DPTModel,
)
from .minigpt4 import (  # noqa: F401
    MiniGPT4Config,
    MiniGPT4ForConditionalGeneration,
)
from .visualglm import (  # noqa: F401
    VisualGLMConfig,
    VisualGLMForConditionalGeneration,
)
from .ernie_vil import (  # noqa: F401
    ErnieViLConfig,
    ErnieViLModel,
)
from .dallebart import (  # noqa: F401
    DalleBartConfig,
    DalleBartForConditionalGeneration,
    DalleBartModel,
)
from .artist import (  # noqa: F401
    ArtistConfig,
    ArtistForConditionalGeneration,
    ArtistModel,
)
from .ernie_gen import (  # noqa: F401
    ErnieGenConfig,
    ErnieGenForGeneration,
    ErnieGenModel,
)
from .clipseg import (  # noqa: F401
    CLIPSegConfig,
    CLIPSegForImageSegmentation,
)
from .semantic_search import (  # noqa: F401
    ErnieCrossEncoder,
    ErnieDualEncoder,
)
from .layoutlmv2 import (  # noqa: F401
    LayoutLMv2Config,
    LayoutLMv2ForTokenClassification,
    LayoutLMv2Model,
)
from .layoutxlm import (  # noqa: F401
    LayoutXLMConfig,
    LayoutXLMForTokenClassification,
    LayoutXLMModel,
)
