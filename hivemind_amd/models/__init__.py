from .albert import AlbertConfig, AlbertForMaskedLM, AlbertLayer, AlbertModel, FusedLayerNorm
from .llama import LlamaConfig, LlamaDecoderLayer, LlamaForCausalLM
