from .sam import SAM_CONFIGS, SamConfig, SamModel
from .dit import DIT_CONFIGS, DiT, DiTConfig
from .deepseek_v3 import DEEPSEEK_V3_CONFIGS, DeepseekV3Config, DeepseekV3ForCausalLM
from .chatglm2 import CHATGLM_CONFIGS, ChatGLMConfig, ChatGLMForConditionalGeneration
from .blip2 import Blip2Config, Blip2ForConditionalGeneration
from .llama import LLAMA_CONFIGS, LlamaConfig, LlamaForCausalLM, llama_flops_per_token
from .bert import BERT_CONFIGS, BertConfig, BertForMaskedLM, BertForSequenceClassification
from .opt import OPT_CONFIGS, OPTConfig, OPTForCausalLM
from .t5 import T5_CONFIGS, T5Config, T5ForConditionalGeneration
from .bloom import BLOOM_CONFIGS, BloomConfig, BloomForCausalLM
from .cohere import COHERE_CONFIGS, CohereConfig, CohereForCausalLM
from .deepseek import DEEPSEEK_CONFIGS, DeepseekConfig, DeepseekForCausalLM
from .gptj import GPTJ_CONFIGS, GPTJConfig, GPTJForCausalLM
from .falcon import FALCON_CONFIGS, FalconConfig, FalconForCausalLM
from .vit import VIT_CONFIGS, ViTConfig, ViTForImageClassification
from .whisper import WHISPER_CONFIGS, WhisperConfig, WhisperForConditionalGeneration

__all__ = ["LlamaConfig", "LlamaForCausalLM", "LLAMA_CONFIGS", "llama_flops_per_token",
           "OPTConfig", "OPTForCausalLM", "OPT_CONFIGS",
           "BertConfig", "BertForMaskedLM", "BertForSequenceClassification", "BERT_CONFIGS",
           "T5Config", "T5ForConditionalGeneration", "T5_CONFIGS",
           "ViTConfig", "ViTForImageClassification", "VIT_CONFIGS",
           "FalconConfig", "FalconForCausalLM", "FALCON_CONFIGS",
           "WhisperConfig", "WhisperForConditionalGeneration", "WHISPER_CONFIGS",
           "BloomConfig", "BloomForCausalLM", "BLOOM_CONFIGS",
           "DeepseekConfig", "DeepseekForCausalLM", "DEEPSEEK_CONFIGS",
           "GPTJConfig", "GPTJForCausalLM", "GPTJ_CONFIGS",
           "CohereConfig", "CohereForCausalLM", "COHERE_CONFIGS"]
