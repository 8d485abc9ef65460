from .llama import (LLAMA_CONFIGS, LlamaConfig,  # noqa: F401
                    LlamaForCausalLM, LlamaModel)
from .mixtral import (MIXTRAL_CONFIGS, MixtralConfig,  # noqa: F401
                      MixtralForCausalLM)
from .gpt2 import GPT2_CONFIGS, GPT2Config, GPT2LMHeadModel  # noqa: F401
from .hf import config_from_hf, load_hf_llama  # noqa: F401
from .bert import (BERT_CONFIGS, BertConfig,  # noqa: F401
                   BertForPreTraining, BertForQuestionAnswering,
                   BertModel)
