from .albert import AlbertConfig, AlbertForMaskedLM, AlbertLayer, AlbertModel, FusedLayerNorm
